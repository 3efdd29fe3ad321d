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
