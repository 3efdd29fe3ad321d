"""Block-sparse attention (reference parity: flashinfer/sparse.py
BlockSparseAttentionWrapper:284). The BSR pattern (indptr/indices over
R x C blocks) is lowered onto the paged prefill kernel: each C-wide block
column becomes a "page" (page_size = C), each R-row block a "request"."""
from __future__ import annotations

from typing import Optional

import torch

from ._lib import get_ext
from .prefill import BatchPrefillWithPagedKVCacheWrapper


class BlockSparseAttentionWrapper:
    def __init__(self, float_workspace_buffer: torch.Tensor, backend: str = "fa2"):
        self._wrapper = BatchPrefillWithPagedKVCacheWrapper(
            float_workspace_buffer, "NHD"
        )
        self.device = float_workspace_buffer.device

    def reset_workspace_buffer(self, float_workspace_buffer, int_workspace_buffer=None):
        self._wrapper.reset_workspace_buffer(float_workspace_buffer)

    def plan(
        self,
        indptr: torch.Tensor,
        indices: torch.Tensor,
        M: int,
        N: int,
        R: int,
        C: int,
        num_qo_heads: int,
        num_kv_heads: int,
        head_dim: int,
        mask: Optional[torch.Tensor] = None,
        packed_mask: Optional[torch.Tensor] = None,
        q_data_type=torch.bfloat16,
        kv_data_type=None,
        o_data_type=None,
        non_blocking: bool = True,
        **kwargs,
    ):
        if mask is not None or packed_mask is not None:
            raise NotImplementedError("per-element masks arrive in a later drop")
        if M % R != 0 or N % C != 0:
            raise ValueError("M/N must be divisible by block sizes R/C")
        n_blockrows = M // R
        qo_indptr = torch.arange(0, M + 1, R, dtype=torch.int32)
        last_page_len = torch.full((n_blockrows,), C, dtype=torch.int32)
        self._shape = (M, N, R, C, num_qo_heads, num_kv_heads, head_dim)
        self._wrapper.plan(
            qo_indptr, indptr, indices, last_page_len,
            num_qo_heads, num_kv_heads, head_dim, C,
            causal=False, q_data_type=q_data_type,
        )

    begin_forward = plan

    def run(self, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
            out=None, lse=None, return_lse: bool = False):
        r"""q: [M, Hq, D]; k/v: [N, Hkv, D] (NHD)."""
        M, N, R, C, Hq, Hkv, D = self._shape
        k4 = k.view(N // C, C, Hkv, D)
        v4 = v.view(N // C, C, Hkv, D)
        return self._wrapper.run(q, (k4, v4), out=out, lse=lse,
                                 return_lse=return_lse)

    forward = run
