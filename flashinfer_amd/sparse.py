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
        if M % R != 0 or N % C != 0:
            raise ValueError("M/N must be divisible by block sizes R/C")
        n_blockrows = M // R
        qo_indptr = torch.arange(0, M + 1, R, dtype=torch.int32)
        last_page_len = torch.full((n_blockrows,), C, dtype=torch.int32)
        self._shape = (M, N, R, C, num_qo_heads, num_kv_heads, head_dim)
        custom_mask = None
        if mask is not None and packed_mask is None:
            # (nnz, R, C) per selected block -> per-request [R, cnt*C] rows
            # (reference sparse.py convert_bsr_mask_layout:170 layout)
            ip = indptr.to("cpu", torch.int64)
            mk = mask.to(torch.bool)
            parts = [
                mk[int(ip[i]):int(ip[i + 1])].transpose(0, 1).reshape(-1)
                for i in range(n_blockrows)
            ]
            custom_mask = torch.cat(parts)
        self._wrapper.plan(
            qo_indptr, indptr, indices, last_page_len,
            num_qo_heads, num_kv_heads, head_dim, C,
            causal=False, q_data_type=q_data_type,
            custom_mask=custom_mask, packed_custom_mask=packed_mask,
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


class VariableBlockSparseAttentionWrapper:
    """Block-sparse attention with variable block sizes and a per-kv-head
    sparse pattern (reference parity: flashinfer/sparse.py
    VariableBlockSparseAttentionWrapper:1193, plan:1300, run:1583).

    Lowering (same shape as the reference's): each (kv_head, row_block)
    becomes one ragged "request" with 1 kv head and ``gqa_group`` query
    heads; the selected column blocks are expanded to token-granular page
    indices (page_size = 1) into the [Hkv*kv_len] flattened cache, and the
    whole batch runs through the paged MFMA prefill kernel in one launch."""

    def __init__(self, float_workspace_buffer: torch.Tensor, backend: str = "fa2"):
        self._wrapper = BatchPrefillWithPagedKVCacheWrapper(
            float_workspace_buffer, "NHD"
        )
        self.device = float_workspace_buffer.device

    def reset_workspace_buffer(self, float_workspace_buffer, int_workspace_buffer=None):
        self._wrapper.reset_workspace_buffer(float_workspace_buffer)

    def plan(
        self,
        block_mask_map: torch.Tensor,   # [Hkv, MB, NB] bool
        block_row_sz: torch.Tensor,     # [Hkv, MB] int
        block_col_sz: torch.Tensor,     # [Hkv, NB] int
        num_qo_heads: int,
        num_kv_heads: int,
        head_dim: int,
        causal: bool = False,
        sm_scale: Optional[float] = None,
        logits_soft_cap: Optional[float] = None,
        q_data_type=torch.bfloat16,
        kv_data_type=None,
        non_blocking: bool = True,
        **kwargs,
    ):
        if num_qo_heads % num_kv_heads != 0:
            raise ValueError("num_qo_heads must be a multiple of num_kv_heads")
        if block_mask_map.shape[0] != num_kv_heads:
            raise ValueError("block_mask_map must have num_kv_heads patterns")
        mm = block_mask_map.to("cpu", torch.bool)
        rsz = block_row_sz.to("cpu", torch.int64)
        csz = block_col_sz.to("cpu", torch.int64)
        H, MB, NB = mm.shape

        # one request per (kv_head, row_block): q rows are contiguous in the
        # [Hkv, qo_len] flattening, so qo_indptr is just the running row count
        qo_indptr = torch.zeros(H * MB + 1, dtype=torch.int32)
        qo_indptr[1:] = torch.cumsum(rsz.reshape(-1), 0).to(torch.int32)

        # expand selected column blocks to token indices in the flattened
        # [Hkv * kv_len] cache (page_size = 1)
        col_off = torch.cumsum(csz, 1) - csz              # [H, NB] start of block c
        head_len = csz.sum(1)                             # [H]
        head_off = torch.cumsum(head_len, 0) - head_len   # [H]
        h_idx, r_idx, c_idx = mm.nonzero(as_tuple=True)
        blk_len = csz[h_idx, c_idx]
        base = head_off[h_idx] + col_off[h_idx, c_idx]
        starts = torch.cumsum(blk_len, 0) - blk_len
        total = int(blk_len.sum())
        within = torch.arange(total, dtype=torch.int64) - torch.repeat_interleave(
            starts, blk_len
        )
        kv_indices = (torch.repeat_interleave(base, blk_len) + within).to(torch.int32)
        # request order must be (h, r) row-major; nonzero() already emits h,r,c
        # in lexicographic order, so per-request runs are contiguous
        row_len = (mm * csz[:, None, :]).sum(-1).reshape(-1)  # [H*MB]
        kv_indptr = torch.zeros(H * MB + 1, dtype=torch.int32)
        kv_indptr[1:] = torch.cumsum(row_len, 0).to(torch.int32)
        last_page_len = torch.ones(H * MB, dtype=torch.int32)

        self._shape = (H, MB, NB, num_qo_heads // num_kv_heads, head_dim)
        self._o_dtype = q_data_type
        self._wrapper.plan(
            qo_indptr, kv_indptr, kv_indices, last_page_len,
            num_qo_heads // num_kv_heads, 1, head_dim, 1,
            causal=causal, sm_scale=sm_scale, logits_soft_cap=logits_soft_cap,
            q_data_type=q_data_type, kv_data_type=kv_data_type,
            non_blocking=non_blocking,
        )

    begin_forward = plan

    def run(self, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
            out=None, lse=None, return_lse: bool = False):
        r"""q: [num_qo_heads, qo_len, D]; k/v: [num_kv_heads, kv_len, D] (HND).
        Returns out of shape [num_qo_heads, qo_len, D]."""
        H, MB, NB, G, D = self._shape
        qo_len = q.shape[1]
        # [H*G, L, D] -> [(H L), G, D] rows grouped per kv head
        q_r = (q.view(H, G, qo_len, D).permute(0, 2, 1, 3)
               .reshape(H * qo_len, G, D).contiguous())
        kv_len = k.shape[1]
        k4 = k.reshape(H * kv_len, 1, 1, D)
        v4 = v.reshape(H * kv_len, 1, 1, D)
        res = self._wrapper.run(q_r, (k4, v4), return_lse=return_lse)
        o_r, lse_r = res if return_lse else (res, None)
        o = (o_r.view(H, qo_len, G, D).permute(0, 2, 1, 3)
             .reshape(H * G, qo_len, D).contiguous())
        if out is not None:
            out.copy_(o)
            o = out
        if return_lse:
            l = (lse_r.view(H, qo_len, G).permute(0, 2, 1)
                 .reshape(H * G, qo_len).contiguous())
            if lse is not None:
                lse.copy_(l)
                l = lse
            return o, l
        return o

    forward = run
