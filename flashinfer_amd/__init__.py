"""flashinfer_amd — an MI355X-native (CDNA4/gfx950) LLM inference kernel
library with FlashInfer's capabilities: plan/run attention wrappers over
paged/ragged KV caches, MLA, cascade attention, state merging, fused
norm/rope/activation/sampling ops, MFMA GEMM (bf16/fp8), fused MoE, and
RCCL-over-xGMI communication helpers.

Hand-written HIP kernels, one backend per op — no CUDA compatibility layer,
no CUTLASS/cuDNN/Triton dispatch.
"""

__version__ = "0.1.0"

from .activation import gelu_and_mul, gelu_tanh_and_mul, silu_and_mul
from .cascade import (
    MultiLevelCascadeAttentionWrapper,
    merge_state,
    merge_state_in_place,
    merge_states,
)
from .sparse import (
    BlockSparseAttentionWrapper,
    VariableBlockSparseAttentionWrapper,
)
from .mla import BatchMLAPagedAttentionWrapper
from .attention import (
    BatchAttention,
    BatchPODWithPagedKVCacheWrapper,
    PODWithPagedKVCacheWrapper,
)
from .mamba import (
    cake_selective_state_update,
    mamba_chunk_scan_combined,
    selective_state_update,
    ssu_checkpoint,
    ssu_rollback,
)
from .green_ctx import split_device_cu_streams, split_device_green_ctx
from .gdn import (
    chunk_gated_delta_rule,
    chunk_kda,
    fused_kda_decode,
    gdn_fused_decode_step,
)
from .fused_moe import cutlass_fused_moe, dsv3_routing, fused_moe, moe_topk_softmax
from .topk import (
    TopKTieBreak,
    top_k,
    top_k_page_table_transform,
    top_k_varlen,
    top_k_ragged_transform,
)
from .quantization import packbits, segment_packbits
from .fp8_quantization import (
    bmm_fp8,
    gemm_fp8_nt_groupwise,
    group_gemm_fp8_nt_groupwise,
    mm_fp8,
    per_block_quant_fp8,
    per_token_group_quant_fp8,
)
from . import comm
from .parallel_attention import ParallelAttention, ring_attention
from .decode import (
    BatchDecodeWithPagedKVCacheWrapper,
    CUDAGraphBatchDecodeWithPagedKVCacheWrapper,
    single_decode_with_kv_cache,
)
from .norm import (
    fused_add_rmsnorm,
    fused_add_rmsnorm_quant,
    fused_rmsnorm_silu,
    fused_qk_rmsnorm_rope,
    gemma_fused_add_rmsnorm,
    gemma_rmsnorm,
    layernorm,
    layernorm_quant,
    rmsnorm,
    rmsnorm_quant,
)
from .prefill import (
    BatchPrefillWithPagedKVCacheWrapper,
    BatchPrefillWithRaggedKVCacheWrapper,
    single_prefill_with_kv_cache,
)
from .gemm import SegmentGEMMWrapper, mm_bf16
from . import sampling
from .sampling import (
    chain_speculative_sampling,
    min_p_sampling_from_probs,
    sampling_from_logits,
    sampling_from_probs,
    softmax,
    top_k_mask_logits,
    top_k_renorm_probs,
    top_k_sampling_from_probs,
    top_k_top_p_sampling_from_logits,
    top_k_top_p_sampling_from_probs,
    top_p_renorm_probs,
    top_p_sampling_from_probs,
)
from .page import append_paged_kv_cache, get_batch_indices_positions, get_seq_lens
from .rope import (
    apply_llama31_rope,
    apply_llama31_rope_inplace,
    apply_llama31_rope_pos_ids,
    apply_llama31_rope_pos_ids_inplace,
    apply_rope,
    apply_rope_inplace,
    apply_rope_pos_ids,
    apply_rope_pos_ids_inplace,
    apply_rope_with_cos_sin_cache,
    apply_rope_with_cos_sin_cache_inplace,
)

__all__ = [  # noqa: F405
    "BatchPrefillWithPagedKVCacheWrapper",
    "BatchPrefillWithRaggedKVCacheWrapper",
    "single_prefill_with_kv_cache",
    "mm_bf16",
    "BatchDecodeWithPagedKVCacheWrapper",
    "CUDAGraphBatchDecodeWithPagedKVCacheWrapper",
    "single_decode_with_kv_cache",
    "merge_state",
    "merge_state_in_place",
    "merge_states",
    "rmsnorm",
    "gemma_rmsnorm",
    "fused_add_rmsnorm",
    "gemma_fused_add_rmsnorm",
    "layernorm",
    "silu_and_mul",
    "gelu_and_mul",
    "gelu_tanh_and_mul",
    "append_paged_kv_cache",
    "get_batch_indices_positions",
    "get_seq_lens",
    "apply_rope",
    "apply_rope_inplace",
    "apply_rope_pos_ids",
    "apply_rope_pos_ids_inplace",
    "apply_llama31_rope",
    "apply_llama31_rope_inplace",
    "apply_llama31_rope_pos_ids",
    "apply_llama31_rope_pos_ids_inplace",
    "apply_rope_with_cos_sin_cache",
    "apply_rope_with_cos_sin_cache_inplace",
]

from . import profiler
from . import concat_ops
from . import logits_processor
from . import msa_ops
from . import testing
from .xqa import xqa, xqa_mla
from . import deep_gemm
from . import moe_ep
from . import trace_apply
from . import collect_env
trace_apply._enable_apply_from_env()
from . import grouped_mm
from .grouped_mm import grouped_mm_bf16, grouped_mm_fp8
from . import dsv3_ops
from . import diffusion_ops
from .mhc import mhc_post, mhc_pre_big_fuse, mhc_pre_big_fuse_with_prenorm
from .concat_ops import concat_mla_k
from . import jit
from . import autotuner
from .autotuner import autotune
from .decode import fast_decode_plan
from .attention import BatchAttentionWithAttentionSinkWrapper
from .cascade import (
    BatchDecodeWithSharedPrefixPagedKVCacheWrapper,
    BatchPrefillWithSharedPrefixPagedKVCacheWrapper,
)
from .mla import BatchDecodeMlaWithPagedKVCacheWrapper
from .gemm import bmm_bf16
from .prefill import single_prefill_with_kv_cache_return_lse
from .page import append_paged_mla_kv_cache
from .gdn import gdn_fused_decode_step_supported
from . import kda
from .kda import (
    RecurrentKDAPrefillWorkspace,
    RecurrentKDAPrefillWrapper,
    packed_kda_decode,
    recurrent_kda,
)
from .fused_moe import (
    ActivationType,
    RoutingMethodType,
    is_gated_activation,
    trtllm_bf16_moe,
    trtllm_bf16_routed_moe,
    trtllm_fp8_block_scale_routed_moe,
    trtllm_fp8_per_tensor_scale_moe,
    trtllm_fp8_per_tensor_scale_routed_moe,
)
from .utils import next_positive_power_of_2
from .norm import (
    fused_dit_gate_residual_layernorm_gamma_beta,
    fused_dit_gate_residual_layernorm_scale_shift,
    fused_dit_residual_layernorm_scale_shift,
)
from .fused_moe import trtllm_fp8_block_scale_moe
