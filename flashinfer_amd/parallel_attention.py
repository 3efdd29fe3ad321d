"""Sequence/context-parallel attention (reference parity:
flashinfer/parallel_attention/ ParallelAttention + ring wrapper
parallel_wrapper.py:216).

Two composable mechanisms over torch.distributed (RCCL on the MI355X node):
  * Ulysses (head scatter/gather all-to-all) — comm/ulysses.py
  * Ring attention: KV rotates around the CP ring with P2P isend/irecv while
    each rank attends its local Q against the visiting KV chunk; partials are
    merged with the LSE merge op (base-2 convention throughout).
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist

from .comm.ulysses import UlyssesCommunicator


def _merge_inplace(v, s, v_new, s_new):
    if v.is_cuda:
        from .cascade import merge_state_in_place

        merge_state_in_place(v, s, v_new, s_new)
    else:  # CPU/gloo reference path
        s_nat, sn_nat = s * math.log(2), s_new * math.log(2)
        m = torch.maximum(s_nat, sn_nat)
        wa, wb = torch.exp(s_nat - m), torch.exp(sn_nat - m)
        tot = wa + wb
        v.copy_(((v.float() * wa[..., None] + v_new.float() * wb[..., None])
                 / tot[..., None]).to(v.dtype))
        s.copy_((m + torch.log(tot)) / math.log(2))


def _local_attn(q, k, v, sm_scale):
    """Attention of q [M,Hq,D] against k/v [L,Hkv,D] -> (out, lse base-2)."""
    if q.is_cuda:
        from .prefill import single_prefill_with_kv_cache

        return single_prefill_with_kv_cache(q, k, v, causal=False,
                                            sm_scale=sm_scale, return_lse=True)
    # CPU reference
    M, Hq, D = q.shape
    L, Hkv, _ = k.shape
    g = Hq // Hkv
    kf = k.float().repeat_interleave(g, dim=1)
    vf = v.float().repeat_interleave(g, dim=1)
    logits = torch.einsum("mhd,lhd->hml", q.float(), kf) * sm_scale
    lse = torch.logsumexp(logits, -1) / math.log(2)  # [H, M]
    p = torch.softmax(logits, dim=-1)
    out = torch.einsum("hml,lhd->mhd", p, vf).to(q.dtype)
    return out, lse.transpose(0, 1).contiguous()


def ring_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    group: Optional[dist.ProcessGroup] = None,
    sm_scale: Optional[float] = None,
) -> torch.Tensor:
    r"""Context-parallel attention: each rank holds a KV shard [L_local] and a
    Q shard; KV shards rotate around the ring, partial states merge by LSE.
    Non-causal (full) attention across the union of all KV shards."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if sm_scale is None:
        sm_scale = 1.0 / math.sqrt(q.shape[-1])
    out, lse = _local_attn(q, k, v, sm_scale)
    out = out.float() if not out.is_cuda else out
    cur_k, cur_v = k.contiguous(), v.contiguous()
    for step in range(1, world):
        nxt = (rank + 1) % world
        prv = (rank - 1) % world
        recv_k = torch.empty_like(cur_k)
        recv_v = torch.empty_like(cur_v)
        reqs = dist.batch_isend_irecv([
            dist.P2POp(dist.isend, cur_k, nxt, group=group),
            dist.P2POp(dist.isend, cur_v, nxt, group=group),
            dist.P2POp(dist.irecv, recv_k, prv, group=group),
            dist.P2POp(dist.irecv, recv_v, prv, group=group),
        ])
        for r in reqs:
            r.wait()
        cur_k, cur_v = recv_k, recv_v
        o_i, lse_i = _local_attn(q, cur_k, cur_v, sm_scale)
        _merge_inplace(out, lse, o_i.to(out.dtype), lse_i)
    return out.to(q.dtype)


class ParallelAttention:
    r"""Composable Ulysses + ring context parallelism (reference
    parallel_attention.py:12). ulysses_group scatters heads; ring_group
    rotates KV."""

    def __init__(self, ulysses_group=None, ring_group=None):
        self.ulysses = (
            UlyssesCommunicator(ulysses_group) if ulysses_group is not None else None
        )
        self.ring_group = ring_group

    def __call__(self, q, k, v, sm_scale=None):
        # q/k/v: [B, S_local, H, D]
        if self.ulysses is not None:
            q = self.ulysses.scatter_heads(q)
            k = self.ulysses.scatter_heads(k)
            v = self.ulysses.scatter_heads(v)
        B, S, H, D = q.shape
        outs = []
        for b in range(B):
            if self.ring_group is not None or dist.is_initialized():
                o = ring_attention(q[b], k[b], v[b], group=self.ring_group,
                                   sm_scale=sm_scale)
            else:
                o, _ = _local_attn(q[b], k[b], v[b],
                                   sm_scale or 1.0 / math.sqrt(D))
            outs.append(o)
        out = torch.stack(outs)
        if self.ulysses is not None:
            out = self.ulysses.gather_heads(out)
        return out
