"""JIT attention-variant customization (reference parity: the
csrc/batch_prefill_customize_config.jinja ADDITIONAL_* injection +
include/flashinfer/attention/variants.cuh:31 / variant_helper.cuh:75
AttentionVariant programming model, and jit/attention/modules.py
gen_customize_batch_prefill_module:1601).

MI355X-native: the user supplies device-code bodies for a logits transform
and/or a logits mask; they compile (hipcc, gfx950) into a variant
instantiation of the SAME prefill tile body the in-tree kernels use
(fi/prefill_body.hpp), loaded as a ctypes module whose entry point the main
binding dispatches through a function pointer — no torch extension rebuild,
no jinja, one content-hashed cached .so per variant.

Example::

    mod = gen_customize_batch_prefill_module(
        "tanh_cap30",
        logits_transform="return 30.f * tanhf(s / 30.f);")
    w = BatchPrefillWithRaggedKVCacheWrapper(ws, jit_module=mod)
    w.plan(...); w.run(q, k, v)   # runs the custom-variant kernel
"""
from __future__ import annotations

import ctypes
from dataclasses import dataclass

from . import gen_jit_spec

_TEMPLATE = r"""
#include "fi/prefill_body.hpp"

namespace fi {

struct UserVariant {
  static constexpr bool kActive = true;
  // s: pre-log2 logit (already sm_scale'd); qo_idx: query position in the
  // request; kv_idx: key position; head: qo head; qo_len/kv_len: request
  // lengths. Return the transformed logit.
  __device__ static float logits_transform(float s, int qo_idx, int64_t kv_idx,
                                           int head, int qo_len,
                                           int64_t kv_len) {
    {LOGITS_TRANSFORM}
  }
  // return false to mask the (qo_idx, kv_idx) position out
  __device__ static bool logits_mask(int qo_idx, int64_t kv_idx, int head,
                                     int qo_len, int64_t kv_len) {
    {LOGITS_MASK}
  }
};

template <typename T, int HD, int CTAQ, bool PAGED, bool CAUSAL>
__global__ __launch_bounds__(CTAQ * 2, ((HD >= 192) ? 1 : 2)) void variant_kernel(
    PrefillParams p) {
  __shared__ char smem[prefill_tile_smem_bytes<T, HD, HD>()];
  const int tile = blockIdx.x;
  if (tile >= p.n_tiles) return;
  prefill_tile_body<T, T, HD, HD, CTAQ, PAGED, CAUSAL, false, UserVariant>(
      p, p.tile_req[tile], p.tile_qstart[tile], blockIdx.y,
      p.tile_kv_chunk ? (int)p.tile_kv_chunk[tile] : -1, smem);
}

template <typename T>
hipError_t variant_dispatch(PrefillParams& p, bool paged, hipStream_t stream) {
  dim3 g(p.n_tiles, p.num_kv_heads), blk(p.cta_q * 2);
#define LV(HD, CQ, PG, CS) \
  hipLaunchKernelGGL((variant_kernel<T, HD, CQ, PG, CS>), g, blk, 0, stream, p)
#define LV2(HD)                                                \
  do {                                                         \
    if (paged) {                                               \
      if (p.causal) { if (p.cta_q == 256) LV(HD, 256, true, true); else LV(HD, 128, true, true); } \
      else { if (p.cta_q == 256) LV(HD, 256, true, false); else LV(HD, 128, true, false); }        \
    } else {                                                   \
      if (p.causal) { if (p.cta_q == 256) LV(HD, 256, false, true); else LV(HD, 128, false, true); } \
      else { if (p.cta_q == 256) LV(HD, 256, false, false); else LV(HD, 128, false, false); }      \
    }                                                          \
  } while (0)
  switch (p.head_dim) {
{HEAD_DIM_CASES}
    default: return hipErrorInvalidValue;
  }
#undef LV2
#undef LV
  return hipGetLastError();
}

}  // namespace fi

extern "C" hipError_t fi_batch_prefill_custom(int dtype, fi::PrefillParams* p,
                                              int paged, hipStream_t stream) {
  if (p->n_tiles == 0) return hipSuccess;
  switch (dtype) {
    case 0: return fi::variant_dispatch<fi::bf16>(*p, paged != 0, stream);
    case 1: return fi::variant_dispatch<fi::fp16>(*p, paged != 0, stream);
  }
  return hipErrorInvalidValue;
}
"""


@dataclass
class AttentionVariantModule:
    name: str
    run_ptr: int  # address of fi_batch_prefill_custom (fn-pointer dispatch)
    _lib: ctypes.CDLL = None


def gen_customize_batch_prefill_module(
    name: str,
    logits_transform: str = "return s;",
    logits_mask: str = "return true;",
    head_dims=(64, 128),
    verbose: bool = False,
) -> AttentionVariantModule:
    """Compile a custom attention variant into a loadable prefill module.

    ``logits_transform`` / ``logits_mask`` are C++ device-function BODIES
    (see the template's signatures). Pass the result as ``jit_module=`` to
    ``BatchPrefillWith{Ragged,Paged}KVCacheWrapper`` (also reachable through
    ``single_prefill_with_kv_cache``'s wrapper path).
    """
    cases = "\n".join(f"    case {hd}: LV2({hd}); break;" for hd in head_dims)
    src = (_TEMPLATE
           .replace("{LOGITS_TRANSFORM}", logits_transform)
           .replace("{LOGITS_MASK}", logits_mask)
           .replace("{HEAD_DIM_CASES}", cases))
    spec = gen_jit_spec(f"prefill_variant_{name}", {"variant.hip": src})
    lib = spec.build_and_load(verbose=verbose)
    fn = lib.fi_batch_prefill_custom
    ptr = ctypes.cast(fn, ctypes.c_void_p).value
    return AttentionVariantModule(name=name, run_ptr=ptr, _lib=lib)
