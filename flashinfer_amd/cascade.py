"""Attention-state merge ops + cascade attention (reference parity:
flashinfer/cascade.py). ``s`` values are base-2 log-sum-exp f32 tensors."""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ._lib import get_ext


def merge_state(
    v_a: torch.Tensor, s_a: torch.Tensor, v_b: torch.Tensor, s_b: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Merge two attention states (V: [n, h, d], S: [n, h])."""
    v = v_a.clone()
    s = s_a.clone().float()
    get_ext().merge_state_in_place(v, s, v_b, s_b.float().contiguous(), None)
    return v, s


def merge_state_in_place(
    v: torch.Tensor, s: torch.Tensor, v_other: torch.Tensor, s_other: torch.Tensor,
    mask: Optional[torch.Tensor] = None,
) -> None:
    get_ext().merge_state_in_place(
        v, s, v_other, s_other.float().contiguous(),
        mask.to(torch.uint8) if mask is not None else None,
    )


def merge_states(v: torch.Tensor, s: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Merge ``num_states`` attention states: v [n, ns, h, d], s [n, ns, h]."""
    n, ns, h, d = v.shape
    v_out = torch.empty((n, h, d), dtype=v.dtype, device=v.device)
    s_out = torch.empty((n, h), dtype=torch.float32, device=v.device)
    get_ext().merge_states(
        v.contiguous().view(n * ns, h, d), s.float().contiguous().view(n * ns, h),
        v_out, s_out, None, ns, n,
    )
    return v_out, s_out


class MultiLevelCascadeAttentionWrapper:
    r"""Shared-prefix hierarchical (cascade) attention over a unified paged KV
    cache (reference parity: flashinfer/cascade.py:226). Level 0 holds KV
    shared by groups of requests (e.g. a common system prompt), deeper levels
    hold progressively more request-specific KV; per-level attention states
    are merged by LSE weight. Causal masking applies to the last level only
    (the unique suffix)."""

    def __init__(self, num_levels: int, float_workspace_buffer: torch.Tensor,
                 kv_layout: str = "NHD", use_cuda_graph: bool = False):
        from .prefill import BatchPrefillWithPagedKVCacheWrapper

        self._num_levels = num_levels
        self._wrappers = [
            BatchPrefillWithPagedKVCacheWrapper(float_workspace_buffer, kv_layout)
            for _ in range(num_levels)
        ]
        self._kv_layout = kv_layout

    def reset_workspace_buffer(self, float_workspace_buffer, int_workspace_buffers=None):
        for w in self._wrappers:
            w.reset_workspace_buffer(float_workspace_buffer)

    def plan(self, qo_indptr_arr, paged_kv_indptr_arr, paged_kv_indices_arr,
             paged_kv_last_page_len, num_qo_heads, num_kv_heads, head_dim,
             page_size, causal: bool = False, pos_encoding_mode: str = "NONE",
             sm_scale=None, window_left: int = -1, logits_soft_cap=None,
             q_data_type=None, kv_data_type=None, **kwargs):
        assert len(qo_indptr_arr) == self._num_levels
        self._qo_indptr_arr = qo_indptr_arr
        for lvl, w in enumerate(self._wrappers):
            is_last = lvl == self._num_levels - 1
            w.plan(
                qo_indptr_arr[lvl], paged_kv_indptr_arr[lvl],
                paged_kv_indices_arr[lvl], paged_kv_last_page_len[lvl],
                num_qo_heads, num_kv_heads, head_dim, page_size,
                causal=causal and is_last, pos_encoding_mode=pos_encoding_mode,
                sm_scale=sm_scale, window_left=window_left if is_last else -1,
                logits_soft_cap=logits_soft_cap, q_data_type=q_data_type,
            )

    begin_forward = plan

    def run(self, q: torch.Tensor, paged_kv_cache, out=None, lse=None):
        v_merged, s_merged = None, None
        for w in self._wrappers:
            v, s = w.run(q, paged_kv_cache, return_lse=True)
            if v_merged is None:
                v_merged, s_merged = v, s
            else:
                merge_state_in_place(v_merged, s_merged, v, s)
        if out is not None:
            out.copy_(v_merged)
            return out
        return v_merged

    forward = run


class BatchDecodeWithSharedPrefixPagedKVCacheWrapper:
    r"""Two-level shared-prefix decode (reference cascade.py older API):
    the shared prefix runs as a single ragged prefill over all queries, the
    per-request suffix as paged decode, and the two states LSE-merge."""

    def __init__(self, float_workspace_buffer: torch.Tensor,
                 kv_layout: str = "NHD"):
        from .decode import BatchDecodeWithPagedKVCacheWrapper

        self._decode = BatchDecodeWithPagedKVCacheWrapper(
            float_workspace_buffer, kv_layout)

    def plan(self, indptr, indices, last_page_len, num_qo_heads, num_kv_heads,
             head_dim, page_size, **kwargs):
        kwargs.setdefault("q_data_type", None)
        qdt = kwargs.pop("q_data_type") or kwargs.pop("data_type", None)
        self._decode.plan(indptr, indices, last_page_len, num_qo_heads,
                          num_kv_heads, head_dim, page_size,
                          q_data_type=qdt or __import__("torch").bfloat16)

    begin_forward = plan

    def forward(self, q, k_shared, v_shared, unique_kv_cache, **kwargs):
        from .prefill import single_prefill_with_kv_cache

        o_s, lse_s = single_prefill_with_kv_cache(
            q, k_shared, v_shared, causal=False, return_lse=True)
        o_u, lse_u = self._decode.run(q, unique_kv_cache, return_lse=True)
        merge_state_in_place(o_s, lse_s, o_u, lse_u)
        return o_s

    run = forward


class BatchPrefillWithSharedPrefixPagedKVCacheWrapper:
    r"""Two-level shared-prefix prefill (reference cascade.py older API)."""

    def __init__(self, float_workspace_buffer: torch.Tensor,
                 kv_layout: str = "NHD"):
        from .prefill import BatchPrefillWithPagedKVCacheWrapper

        self._prefill = BatchPrefillWithPagedKVCacheWrapper(
            float_workspace_buffer, kv_layout)

    def plan(self, qo_indptr, paged_kv_indptr, paged_kv_indices,
             paged_kv_last_page_len, num_qo_heads, num_kv_heads, head_dim,
             page_size, causal: bool = True, **kwargs):
        self._prefill.plan(qo_indptr, paged_kv_indptr, paged_kv_indices,
                           paged_kv_last_page_len, num_qo_heads, num_kv_heads,
                           head_dim, page_size, causal=causal)

    begin_forward = plan

    def forward(self, q, k_shared, v_shared, unique_kv_cache,
                causal: bool = True, **kwargs):
        from .prefill import single_prefill_with_kv_cache

        o_s, lse_s = single_prefill_with_kv_cache(
            q, k_shared, v_shared, causal=False, return_lse=True)
        o_u, lse_u = self._prefill.run(q, unique_kv_cache, return_lse=True)
        merge_state_in_place(o_s, lse_s, o_u, lse_u)
        return o_s

    run = forward
