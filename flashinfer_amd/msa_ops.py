"""Minimax Sparse Attention (MSA) ops (reference parity:
flashinfer/msa_ops/__init__.py — msa_proxy_score (proxy_score.py:267),
msa_topk_select (sparse_topk_select.py:79), msa_sparse_attention
(sparse_prefill.py:35), msa_sparse_decode_attention (sparse_decode.py:173)).

MI355X lowering: the proxy pass is a plain QK^T GEMM + per-128-block amax —
exactly the shape hipBLASLt is for, streamed in KV chunks so nothing
quadratic materializes; block selection runs on the threshold-search top-k
kernel (ragged per-token valid counts, forced sink/local blocks); the sparse
attention itself expands the selected blocks to token-granular page indices
and runs the one CDNA4 MFMA prefill kernel (page_size=1, one request per
(kv_head, query token))."""
from __future__ import annotations

import math
from typing import Optional, Tuple, Union

import torch

from ._lib import get_ext
from .prefill import BatchPrefillWithPagedKVCacheWrapper

BLOCK = 128
SUPPORTS_PACKED_KV = True


def _flat_kv_from_pages(pages: torch.Tensor, page_table: torch.Tensor,
                        seqused_k: torch.Tensor):
    """[num_pages, Hkv, 128, D] + per-batch page rows -> per-seq flat KV list."""
    out, cu = [], [0]
    for b in range(seqused_k.numel()):
        L = int(seqused_k[b])
        npg = (L + BLOCK - 1) // BLOCK
        rows = pages[page_table[b, :npg].long()]        # [npg, Hkv, 128, D]
        flat = rows.permute(0, 2, 1, 3).reshape(npg * BLOCK, *rows.shape[1:4:2])
        out.append(flat[:L])
        cu.append(cu[-1] + L)
    return torch.cat(out), torch.tensor(cu, dtype=torch.int32)


def msa_proxy_score(
    q: torch.Tensor,                      # [total_q, Hq, 128]
    k: torch.Tensor,                      # [total_k, Hkv, 128] or paged
    cu_seqlens_q: torch.Tensor,
    cu_seqlens_k: Optional[torch.Tensor] = None,
    causal: bool = True,
    page_table: Optional[torch.Tensor] = None,
    seqused_k: Optional[torch.Tensor] = None,
    max_seqlen_q: Optional[int] = None,
    max_k_tiles: Optional[int] = None,
    output: Optional[torch.Tensor] = None,
    reduce_heads: bool = False,
    q_offset: Union[int, torch.Tensor, None] = None,
) -> torch.Tensor:
    r"""Per-KV-block max of the unscaled causally-masked QK^T logits:
    ``max_score[h, t, q]`` (float32, invalid blocks = -inf)."""
    if k.dim() == 4:
        if page_table is None or seqused_k is None:
            raise ValueError("paged K requires page_table and seqused_k")
        k, cu_seqlens_k = _flat_kv_from_pages(k, page_table, seqused_k)
    if cu_seqlens_k is None:
        raise ValueError("flat K requires cu_seqlens_k")
    total_q, Hq, D = q.shape
    Hkv = k.shape[1]
    G = Hq // Hkv
    cq = cu_seqlens_q.to("cpu", torch.int64).tolist()
    ck = cu_seqlens_k.to("cpu", torch.int64).tolist()
    B = len(cq) - 1
    if max_k_tiles is None:
        max_k_tiles = max(
            (ck[b + 1] - ck[b] + BLOCK - 1) // BLOCK for b in range(B))
    Ho = 1 if reduce_heads else Hq
    if output is None:
        output = torch.full((Ho, max_k_tiles, total_q), float("-inf"),
                            dtype=torch.float32, device=q.device)
    else:
        output.fill_(float("-inf"))
    qf = q.float()
    for b in range(B):
        Lq, Lk = cq[b + 1] - cq[b], ck[b + 1] - ck[b]
        if Lq == 0 or Lk == 0:
            continue
        qb = qf[cq[b]:cq[b + 1]].permute(1, 0, 2)           # [Hq, Lq, D]
        kb = k[ck[b]:ck[b + 1]].float().permute(1, 2, 0)    # [Hkv, D, Lk]
        off = q_offset if isinstance(q_offset, int) else 0
        qpos = torch.arange(Lq, device=q.device) + (Lk - Lq) + off
        for c0 in range(0, Lk, 32 * BLOCK):                  # stream KV chunks
            c1 = min(c0 + 32 * BLOCK, Lk)
            kv_chunk = kb[:, :, c0:c1]
            # [Hkv, G, Lq, D] @ [Hkv, 1, D, C] -> [Hq, Lq, C]
            s = torch.matmul(qb.view(Hkv, G, Lq, D),
                             kv_chunk.unsqueeze(1)).view(Hq, Lq, c1 - c0)
            if causal:
                kpos = torch.arange(c0, c1, device=q.device)
                s = s.masked_fill(kpos[None, None, :] > qpos[None, :, None],
                                  float("-inf"))
            pad = (-(c1 - c0)) % BLOCK
            if pad:
                s = torch.nn.functional.pad(s, (0, pad), value=float("-inf"))
            blk = s.view(Hq, Lq, -1, BLOCK).amax(-1)        # [Hq, Lq, nblk]
            if reduce_heads:
                blk = blk.amax(0, keepdim=True)
            output[:, c0 // BLOCK: c0 // BLOCK + blk.shape[2],
                   cq[b]:cq[b + 1]] = blk.permute(0, 2, 1)
    return output


def msa_topk_select(
    max_score: torch.Tensor,              # [H, T, Q] f32
    topk: int,
    num_valid_pages: Union[int, torch.Tensor, None] = None,
    output: Optional[torch.Tensor] = None,
    force_begin_blocks: int = 0,
    force_end_blocks: int = 0,
) -> torch.Tensor:
    r"""Top-``topk`` KV blocks per (query token, head): ascending int32
    indices, -1 tail-padded; sink/local blocks force-included."""
    from .topk import top_k_ragged_transform

    H, T, Q = max_score.shape
    dev = max_score.device
    x = max_score.permute(0, 2, 1).reshape(H * Q, T).contiguous()
    if num_valid_pages is None:
        valid = torch.full((Q,), T, dtype=torch.int32, device=dev)
    elif isinstance(num_valid_pages, int):
        valid = torch.full((Q,), min(num_valid_pages, T), dtype=torch.int32,
                           device=dev)
    else:
        valid = num_valid_pages.to(dev, torch.int32).clamp_(0, T)
    if force_begin_blocks or force_end_blocks:
        x = x.clone()
        col = torch.arange(T, device=dev)
        vex = valid.repeat(H).unsqueeze(1)
        forced = (col[None, :] < force_begin_blocks) | (
            (col[None, :] >= vex - force_end_blocks) & (col[None, :] < vex))
        x.masked_fill_(forced, float("inf"))
    lengths = valid.repeat(H)
    zero_off = torch.zeros(H * Q, dtype=torch.int32, device=dev)
    idx = top_k_ragged_transform(x, zero_off, lengths, topk)   # [H*Q, topk]
    # ascending order with -1 at the tail
    key = torch.where(idx < 0, torch.full_like(idx, 1 << 30), idx)
    idx = key.sort(-1).values
    idx[idx == 1 << 30] = -1
    idx = idx.view(H, Q, topk).permute(1, 0, 2).contiguous()
    if output is not None:
        output.copy_(idx)
        return output
    return idx


def _sparse_run(q, k, v, q2k_indices, qpos, seq_of_q, cu_k, causal,
                softmax_scale, return_softmax_lse):
    """Shared lowering: one request per (kv_head, q token), page_size=1."""
    total_q, Hq, D = q.shape
    Hkv = k.shape[1]
    G = Hq // Hkv
    topk = q2k_indices.shape[2]
    dev = q.device
    # expand blocks -> token indices in the [Hkv * total_k] flattened cache
    cu_k_d = cu_k.to(dev, torch.int64)
    seq_q = seq_of_q.to(dev, torch.int64)
    blocks = q2k_indices.to(dev, torch.int64)                   # [Hkv, Q, topk]
    base = cu_k_d[seq_q][None, :, None] + blocks * BLOCK        # global token
    within = torch.arange(BLOCK, device=dev)
    tok = base[..., None] + within                              # [Hkv,Q,topk,128]
    seq_end = cu_k_d[seq_q + 1][None, :, None, None]
    limit = torch.minimum(
        seq_end,
        (cu_k_d[seq_q][None, :] + qpos.to(dev)[None, :] + 1)[..., None, None]
        if causal else seq_end)
    ok = (blocks[..., None] >= 0) & (tok < limit)
    total_k = k.shape[0]
    head_off = (torch.arange(Hkv, device=dev) * total_k)[:, None, None, None]
    tok = tok + head_off
    flat_ok = ok.reshape(Hkv * total_q, topk * BLOCK)
    flat_tok = tok.reshape(Hkv * total_q, topk * BLOCK)
    counts = flat_ok.sum(-1).to(torch.int32)
    kv_indices = flat_tok[flat_ok].to(torch.int32)
    kv_indptr = torch.zeros(Hkv * total_q + 1, dtype=torch.int32, device=dev)
    kv_indptr[1:] = counts.cumsum(0)
    qo_indptr = torch.arange(0, Hkv * total_q + 1, dtype=torch.int32)
    last_page_len = torch.ones(Hkv * total_q, dtype=torch.int32)

    q_r = (q.view(total_q, Hkv, G, D).permute(1, 0, 2, 3)
           .reshape(Hkv * total_q, G, D).contiguous())
    k4 = k.permute(1, 0, 2).reshape(Hkv * total_k, 1, 1, D).contiguous()
    v4 = v.permute(1, 0, 2).reshape(Hkv * total_k, 1, 1, D).contiguous()

    ws = torch.empty(32 << 20, dtype=torch.uint8, device=dev)
    w = BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, last_page_len, G, 1, D, 1,
           causal=False, sm_scale=softmax_scale)
    res = w.run(q_r, (k4, v4), return_lse=return_softmax_lse)
    o_r, lse_r = res if return_softmax_lse else (res, None)
    out = (o_r.view(Hkv, total_q, G, D).permute(1, 0, 2, 3)
           .reshape(total_q, Hq, D).contiguous())
    if return_softmax_lse:
        lse = (lse_r.view(Hkv, total_q, G).permute(1, 0, 2)
               .reshape(total_q, Hq).contiguous()) * math.log(2.0)
        return out, lse
    return out


def msa_sparse_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    q2k_indices: torch.Tensor,            # [Hkv, total_q, topk] int32
    cu_seqlens_q: torch.Tensor,
    cu_seqlens_k: Optional[torch.Tensor] = None,
    causal: bool = False,
    softmax_scale: Optional[float] = None,
    page_table: Optional[torch.Tensor] = None,
    seqused_k: Optional[torch.Tensor] = None,
    return_softmax_lse: bool = False,
    k_scale=None, v_scale=None, k_global_scale=None, v_global_scale=None,
    q_offset=None, return_temperature_lse: bool = False,
    lse_temperature_scale: float = 1.0,
):
    r"""Sparse prefill: each query attends only its selected 128-token KV
    blocks. Returns out [total_q, Hq, D] (+ natural-log LSE)."""
    if k.dim() == 4:
        k, cu_seqlens_k = _flat_kv_from_pages(k, page_table, seqused_k)
        v, _ = _flat_kv_from_pages(v, page_table, seqused_k)
    total_q = q.shape[0]
    if softmax_scale is None:
        softmax_scale = q.shape[-1] ** -0.5
    cq = cu_seqlens_q.to("cpu", torch.int64)
    ck = cu_seqlens_k.to("cpu", torch.int64)
    seq_of_q = torch.repeat_interleave(
        torch.arange(cq.numel() - 1), cq[1:] - cq[:-1])
    lens_q = (cq[1:] - cq[:-1])[seq_of_q]
    lens_k = (ck[1:] - ck[:-1])[seq_of_q]
    pos_in = torch.arange(total_q) - cq[seq_of_q]
    qpos = pos_in + (lens_k - lens_q)                    # right-aligned
    return _sparse_run(q, k, v, q2k_indices, qpos, seq_of_q, ck, causal,
                       softmax_scale, return_softmax_lse)


def msa_sparse_decode_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    q2k_indices: torch.Tensor,
    page_table: Optional[torch.Tensor] = None,
    seqused_k: Optional[torch.Tensor] = None,
    cu_seqlens_k: Optional[torch.Tensor] = None,
    cu_seqlens_q: Optional[torch.Tensor] = None,
    softmax_scale: Optional[float] = None,
    return_softmax_lse: bool = False,
    k_scale=None, v_scale=None, k_global_scale=None, v_global_scale=None,
    q_offset=None,
):
    r"""Sparse decode: one (or a few MTP) query tokens per sequence attend
    their selected KV blocks (full blocks, no causal mask)."""
    total_q = q.shape[0]
    if cu_seqlens_q is None:
        cu_seqlens_q = torch.arange(total_q + 1, dtype=torch.int32)
    return msa_sparse_attention(
        q, k, v, q2k_indices, cu_seqlens_q, cu_seqlens_k, causal=False,
        softmax_scale=softmax_scale, page_table=page_table,
        seqused_k=seqused_k, return_softmax_lse=return_softmax_lse)
