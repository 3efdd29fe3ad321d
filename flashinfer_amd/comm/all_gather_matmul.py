"""All-gather + matmul overlap (reference parity:
flashinfer/comm/all_gather_matmul/all_gather_matmul.py:64 — push-wait AG
matmul). MI355X form: the local shard's GEMM issues immediately while the
RCCL all-gather runs (async_op); remote shards multiply as soon as the
gather lands — compute hides under the 7-link xGMI transfer instead of a
signal-polling kernel."""
from __future__ import annotations

import torch
import torch.distributed as dist


def all_gather_matmul(
    inp: torch.Tensor,       # [m_local, K]
    w: torch.Tensor,         # [K, N]
    group=None,
    *,
    verbose: bool = False,
) -> torch.Tensor:
    r"""Returns ``all_gather(inp) @ w`` of shape [world * m_local, N]."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    m, K = inp.shape
    N = w.shape[1]
    ag = torch.empty(world * m, K, dtype=inp.dtype, device=inp.device)
    if dist.get_backend(group) == "gloo":
        chunks = list(ag.chunk(world, dim=0))
        work = dist.all_gather(chunks, inp.contiguous(), group=group,
                               async_op=True)
    else:
        work = dist.all_gather_into_tensor(ag, inp.contiguous(), group=group,
                                           async_op=True)
    out = torch.empty(world * m, N, dtype=inp.dtype, device=inp.device)
    # local shard computes while the gather is in flight
    torch.matmul(inp, w, out=out[rank * m:(rank + 1) * m])
    work.wait()
    for r in range(world):
        if r != rank:
            torch.matmul(ag[r * m:(r + 1) * m], w,
                         out=out[r * m:(r + 1) * m])
    return out
