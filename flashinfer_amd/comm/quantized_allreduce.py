"""fp8-compressed two-shot allreduce (reference parity:
flashinfer/comm/quantized_allreduce.py quantized_all_reduce:478). The
payload crosses xGMI as e4m3 bytes + per-group f32 scales — halving the
per-link bytes of the reduce-scatter and all-gather phases — with the
reduction itself in f32. RCCL all_to_all/all_gather data plane (the
symmetric-memory push kernel is a later drop)."""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

SCALE_GROUP_DEFAULT = 256
_FP8_MAX = 448.0


def _quant(x: torch.Tensor, scale_group: int):
    g = x.reshape(-1, scale_group).float()
    scale = g.abs().amax(-1, keepdim=True).clamp(min=1e-10) / _FP8_MAX
    q = (g / scale).clamp(-_FP8_MAX, _FP8_MAX).to(torch.float8_e4m3fn)
    return q.view(torch.uint8).reshape(-1), scale.reshape(-1)


def _dequant(q: torch.Tensor, scale: torch.Tensor, scale_group: int):
    f = q.view(torch.float8_e4m3fn).float().reshape(-1, scale_group)
    return f * scale.reshape(-1, 1)


def _a2a(x: torch.Tensor, group):
    world = dist.get_world_size(group)
    if dist.get_backend(group) == "gloo":
        # gloo has no alltoall: emulate by gathering and slicing own shard
        rank = dist.get_rank(group)
        gathered = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(gathered, x.contiguous(), group=group)
        return torch.cat([g.chunk(world)[rank] for g in gathered])
    out = torch.empty_like(x)
    dist.all_to_all_single(out, x, group=group)
    return out


def quantized_all_reduce(
    inp: torch.Tensor,
    group=None,
    *,
    scale_group: int = SCALE_GROUP_DEFAULT,
    block_size: Optional[int] = None,
    num_warps: Optional[int] = None,
    max_num_blocks: Optional[int] = None,
    p2p_phase3: Optional[bool] = None,
    output: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""Two-shot allreduce with fp8 payloads: quantize -> all-to-all shards
    -> dequantized f32 partial sums -> requantize -> all-gather -> dequant.
    ``inp.numel()`` must divide by ``world * scale_group``."""
    world = dist.get_world_size(group)
    n = inp.numel()
    if n % (world * scale_group) != 0:
        raise ValueError("numel must divide by world * scale_group")
    shard = n // world
    q, s = _quant(inp, scale_group)
    # shot 1: exchange shards; rank r receives every rank's shard r
    qr = _a2a(q, group)
    sr = _a2a(s, group)
    # dequantized f32 reduction of this rank's shard across world sources
    parts = _dequant(qr, sr, scale_group).reshape(world, shard // scale_group,
                                                  scale_group)
    red = parts.sum(0).reshape(-1)
    # shot 2: gather the reduced shards
    q2, s2 = _quant(red, scale_group)
    qg = torch.empty(world * q2.numel(), dtype=torch.uint8, device=inp.device)
    sg = torch.empty(world * s2.numel(), dtype=torch.float32, device=inp.device)
    if dist.get_backend(group) == "gloo":
        dist.all_gather(list(qg.chunk(world)), q2, group=group)
        dist.all_gather(list(sg.chunk(world)), s2, group=group)
    else:
        dist.all_gather_into_tensor(qg, q2, group=group)
        dist.all_gather_into_tensor(sg, s2, group=group)
    res = _dequant(qg, sg, scale_group).reshape(inp.shape).to(inp.dtype)
    if output is not None:
        output.copy_(res)
        return output
    return res


def reset() -> None:
    """Reference-API parity: drop cached comm buffers (none are cached in
    the RCCL data plane)."""
