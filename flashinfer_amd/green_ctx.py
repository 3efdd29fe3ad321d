"""CU-partitioned streams (reference parity: flashinfer/green_ctx.py
split_device_green_ctx — SM-partitioned streams used to colocate
communication with compute). The MI355X-native mechanism is a CU-masked HIP
stream (hipExtStreamCreateWithCUMask): each returned stream is restricted to
a disjoint set of CUs, so a comm kernel pinned to a few CUs cannot starve
the compute stream."""
from __future__ import annotations

import ctypes
from typing import List, Tuple

import torch

_hip = None


def _lib():
    global _hip
    if _hip is None:
        _hip = ctypes.CDLL("libamdhip64.so")
    return _hip


def split_device_cu_streams(
    device: torch.device, groups: List[int]
) -> Tuple[List[torch.cuda.Stream], List[int]]:
    r"""Partition the device's CUs into len(groups) disjoint sets of the given
    sizes and return one CU-masked stream per set (plus the actual CU counts).

    Example: ``split_device_cu_streams(dev, [224, 32])`` gives a big compute
    stream and a 32-CU comm stream.
    """
    props = torch.cuda.get_device_properties(device)
    total = props.multi_processor_count
    if sum(groups) > total:
        raise ValueError(f"requested {sum(groups)} CUs > {total}")
    lib = _lib()
    streams = []
    counts = []
    cu0 = 0
    for n in groups:
        words = (total + 31) // 32
        mask = [0] * words
        for cu in range(cu0, cu0 + n):
            mask[cu // 32] |= 1 << (cu % 32)
        arr = (ctypes.c_uint32 * words)(*mask)
        sp = ctypes.c_void_p()
        rc = lib.hipExtStreamCreateWithCUMask(ctypes.byref(sp), words, arr)
        if rc != 0:
            raise RuntimeError(f"hipExtStreamCreateWithCUMask failed: {rc}")
        streams.append(
            torch.cuda.Stream(stream_ptr=sp.value, device=device)
        )
        counts.append(n)
        cu0 += n
    return streams, counts


# reference-compatible alias
split_device_green_ctx = split_device_cu_streams
