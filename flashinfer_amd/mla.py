"""DeepSeek MLA paged attention wrapper (reference parity:
flashinfer/mla/_core.py BatchMLAPagedAttentionWrapper:1963, plan:2134,
run:2267). Matrix-absorption form: q_nope/ckv share the 512-d compressed
space, q_pe/kpe the 64-d rope space; one hand-written CDNA4 kernel
(csrc/attention/mla_decode.hip) serves decode and incremental prefill."""
from __future__ import annotations

import math
from typing import Optional

import torch

from ._lib import get_ext
from .utils import WorkspaceAllocator, ceil_div


class BatchMLAPagedAttentionWrapper:
    def __init__(
        self,
        float_workspace_buffer: torch.Tensor,
        use_cuda_graph: bool = False,
        qo_indptr: Optional[torch.Tensor] = None,
        kv_indptr: Optional[torch.Tensor] = None,
        kv_indices: Optional[torch.Tensor] = None,
        kv_len_arr: Optional[torch.Tensor] = None,
        backend: str = "fa2",
    ) -> None:
        self._float_workspace_buffer = float_workspace_buffer
        self.device = float_workspace_buffer.device
        self._use_cuda_graph = use_cuda_graph
        self._plan_info = None

    def plan(
        self,
        qo_indptr: torch.Tensor,
        kv_indptr: torch.Tensor,
        kv_indices: torch.Tensor,
        kv_len_arr: torch.Tensor,
        num_heads: int,
        head_dim_ckv: int,
        head_dim_kpe: int,
        page_size: int,
        causal: bool,
        sm_scale: float,
        q_data_type: torch.dtype = torch.bfloat16,
        kv_data_type: Optional[torch.dtype] = None,
        use_profiler: bool = False,
        **kwargs,
    ) -> None:
        if head_dim_ckv != 512 or head_dim_kpe != 64:
            raise NotImplementedError("MLA kernel is specialized for 512+64")
        qi = qo_indptr.to("cpu", torch.int64)
        kv_lens = kv_len_arr.to("cpu", torch.int64).tolist()
        qo_lens = (qi[1:] - qi[:-1]).tolist()
        nnz = int(qi[-1])

        # split-KV chunking: target ONE workgroup per CU (the kernel is 8
        # waves at launch_bounds(512,1) — a CU hosts exactly one WG, so
        # >256 items dispatch in extra rounds; the decode-route sweep
        # measured 2 rounds ~25% slower — profiles/README r02 addendum 6)
        total_rows = nnz * num_heads
        row_tiles = sum(ceil_div(L * num_heads, 64) for L in qo_lens)
        target_items = 256
        chunks_per = max(1, target_items // max(1, row_tiles))
        max_kv = max(kv_lens) if kv_lens else 1
        chunk = max(page_size, 128, math.ceil(max_kv / chunks_per))
        chunk = ceil_div(chunk, page_size) * page_size
        max_chunks = max(1, ceil_div(max_kv, chunk))

        tile_req, tile_row0, tile_chunk = [], [], []
        for b, (qlen, kvlen) in enumerate(zip(qo_lens, kv_lens)):
            rows = qlen * num_heads
            nch = max(1, ceil_div(kvlen, chunk))
            for r0 in range(0, max(rows, 1), 64):
                for c in range(nch):
                    tile_req.append(b)
                    tile_row0.append(r0)
                    tile_chunk.append(c)
        n_items = len(tile_req)
        meta = torch.tensor(tile_req + tile_row0 + tile_chunk, dtype=torch.int32).to(
            self.device, non_blocking=True
        )
        self._tile_req = meta[:n_items]
        self._tile_row0 = meta[n_items : 2 * n_items]
        self._tile_chunk = meta[2 * n_items :]
        self._qo_indptr_d = qo_indptr.to(self.device, torch.int32, non_blocking=True)
        self._kv_indptr_d = kv_indptr.to(self.device, torch.int32, non_blocking=True)
        self._kv_indices_d = kv_indices.to(self.device, torch.int32, non_blocking=True)
        # derive last_page_len from kv_len
        lpl = ((kv_len_arr.to(torch.int64) - 1) % page_size + 1).to(torch.int32)
        self._kv_last_page_len_d = lpl.to(self.device, non_blocking=True)

        alloc = WorkspaceAllocator(self._float_workspace_buffer)
        # partials stored in the q dtype (bf16/f16): halves the split-KV
        # round-trip traffic that bounds the small-batch kernel
        self._tmp_v = alloc.alloc(
            total_rows * max_chunks * 512 * 2, q_data_type,
            (total_rows * max_chunks, 1, 512),
        )
        self._tmp_s = alloc.alloc(
            total_rows * max_chunks * 4, torch.float32, (total_rows * max_chunks, 1)
        )
        # slots the kernel never writes (rows with < max_chunks chunks) must
        # merge as empty; the kernel always (re)writes every valid slot, so
        # one fill at plan time covers every run
        self._tmp_s.fill_(float("-inf"))
        self._plan_info = dict(
            num_heads=num_heads, chunk=chunk, max_chunks=max_chunks,
            causal=causal, sm_scale=sm_scale, total_rows=total_rows, nnz=nnz,
        )

    begin_forward = plan

    def run(
        self,
        q_nope: torch.Tensor,
        q_pe: torch.Tensor,
        ckv_cache: torch.Tensor,
        kpe_cache: torch.Tensor,
        out: Optional[torch.Tensor] = None,
        lse: Optional[torch.Tensor] = None,
        return_lse: bool = False,
        ckv_scale: float = 1.0,
        kpe_scale: float = 1.0,
        **kwargs,
    ):
        pi = self._plan_info
        if pi is None:
            raise RuntimeError("must call plan() before run()")
        get_ext().mla_run(
            q_nope, q_pe, ckv_cache, kpe_cache,
            self._qo_indptr_d, self._kv_indices_d, self._kv_indptr_d,
            self._kv_last_page_len_d, self._tile_req, self._tile_row0,
            self._tile_chunk, pi["chunk"], pi["max_chunks"],
            self._tmp_v, self._tmp_s, pi["sm_scale"], pi["causal"],
            ckv_scale, kpe_scale,
        )
        nnz, H = q_nope.shape[0], pi["num_heads"]
        if out is None:
            out = torch.empty(nnz, H, 512, dtype=q_nope.dtype, device=q_nope.device)
        if return_lse and lse is None:
            lse = torch.empty(nnz, H, dtype=torch.float32, device=q_nope.device)
        # merge chunks: positions are (token, head) rows with "1 head", d=512
        get_ext().merge_states(
            self._tmp_v, self._tmp_s, out.view(nnz * H, 1, 512),
            lse.view(nnz * H, 1) if return_lse else None,
            None, pi["max_chunks"], nnz * H,
        )
        return (out, lse) if return_lse else out

    forward = run

    def end_forward(self):
        pass


# reference decode.py:2406 exposes the MLA decode wrapper under this name;
# the plan/run contract here follows BatchMLAPagedAttentionWrapper
BatchDecodeMlaWithPagedKVCacheWrapper = BatchMLAPagedAttentionWrapper
