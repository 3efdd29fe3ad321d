"""Tensor-parallel allreduce with fused epilogues (reference parity:
flashinfer/comm/allreduce.py + trtllm_ar.py patterns).

MI355X design: the data plane is RCCL over xGMI (7 direct links/GPU — RCCL's
allreduce already engages them); the fusion (residual add + RMSNorm [+ quant])
runs as ONE local kernel afterwards instead of being stitched into the
collective. A hipIpc one-shot AR kernel for small messages is a planned
follow-up; this module is the correctness-complete path and the API surface.

Patterns (reference trtllm_ar.py:68 AllReduceFusionPattern):
  kAllReduce               : out = allreduce(x)
  kARResidualRMSNorm       : residual += AR(x); out = rmsnorm(residual) * w
"""
from __future__ import annotations

from enum import IntEnum
from typing import Optional

import torch
import torch.distributed as dist


class AllReduceFusionPattern(IntEnum):
    kAllReduce = 0
    kARResidualRMSNorm = 1


def allreduce(x: torch.Tensor, group: Optional[dist.ProcessGroup] = None):
    dist.all_reduce(x, group=group)
    return x


def allreduce_fusion(
    x: torch.Tensor,
    residual: Optional[torch.Tensor] = None,
    rms_weight: Optional[torch.Tensor] = None,
    eps: float = 1e-6,
    pattern: AllReduceFusionPattern = AllReduceFusionPattern.kAllReduce,
    group: Optional[dist.ProcessGroup] = None,
):
    r"""Allreduce x across the TP group, then apply the fused epilogue.

    kARResidualRMSNorm: in-place ``residual += AR(x); x = rmsnorm(residual)``
    (returns (norm_out, residual) views of x/residual).
    """
    dist.all_reduce(x, group=group)
    if pattern == AllReduceFusionPattern.kAllReduce:
        return x
    if pattern == AllReduceFusionPattern.kARResidualRMSNorm:
        if residual is None or rms_weight is None:
            raise ValueError("pattern requires residual and rms_weight")
        if x.is_cuda:
            from ..norm import fused_add_rmsnorm

            fused_add_rmsnorm(x, residual, rms_weight, eps)
        else:  # CPU (gloo) reference path for tests
            residual += x
            rf = residual.float()
            rms = torch.rsqrt(rf.pow(2).mean(-1, keepdim=True) + eps)
            x.copy_((rf * rms * rms_weight.float()).to(x.dtype))
        return x, residual
    raise ValueError(f"unknown pattern {pattern}")
