"""hipIpc buffer plumbing (reference parity: flashinfer/comm/cuda_ipc.py —
CudaRTLibrary ctypes wrapper + create_shared_buffer/free_shared_buffer).
The host driver on this pool only supports dmabuf IPC, so keep
``HSA_ENABLE_IPC_MODE_LEGACY=0`` exported (as the image does) for
hipIpcGetMemHandle to work across processes."""
from __future__ import annotations

import ctypes
from typing import List, Optional

import torch

HIP_IPC_HANDLE_SIZE = 64


class hipIpcMemHandle_t(ctypes.Structure):
    _fields_ = [("reserved", ctypes.c_byte * HIP_IPC_HANDLE_SIZE)]

    def bytes_(self) -> bytes:
        return bytes(self.reserved)

    @staticmethod
    def from_bytes(b: bytes) -> "hipIpcMemHandle_t":
        h = hipIpcMemHandle_t()
        ctypes.memmove(h.reserved, b, HIP_IPC_HANDLE_SIZE)
        return h


class HipRTLibrary:
    """ctypes surface of the hip runtime used for IPC buffer lifecycle
    (mirrors the reference's CudaRTLibrary role)."""

    def __init__(self, path: str = "libamdhip64.so"):
        self.lib = ctypes.CDLL(path)
        self.lib.hipMalloc.argtypes = [ctypes.POINTER(ctypes.c_void_p),
                                       ctypes.c_size_t]
        self.lib.hipExtMallocWithFlags.argtypes = [
            ctypes.POINTER(ctypes.c_void_p), ctypes.c_size_t, ctypes.c_uint]
        self.lib.hipFree.argtypes = [ctypes.c_void_p]
        self.lib.hipMemset.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                       ctypes.c_size_t]
        self.lib.hipMemcpy.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_size_t, ctypes.c_int]
        self.lib.hipIpcGetMemHandle.argtypes = [
            ctypes.POINTER(hipIpcMemHandle_t), ctypes.c_void_p]
        self.lib.hipIpcOpenMemHandle.argtypes = [
            ctypes.POINTER(ctypes.c_void_p), hipIpcMemHandle_t, ctypes.c_uint]
        self.lib.hipIpcCloseMemHandle.argtypes = [ctypes.c_void_p]
        self.lib.hipSetDevice.argtypes = [ctypes.c_int]
        self.lib.hipDeviceSynchronize.argtypes = []

    def _check(self, code: int, fn: str):
        if code != 0:
            self.lib.hipGetErrorString.restype = ctypes.c_char_p
            msg = self.lib.hipGetErrorString(code).decode()
            raise RuntimeError(f"{fn}: {msg} ({code})")

    def hipSetDevice(self, dev: int):
        self._check(self.lib.hipSetDevice(dev), "hipSetDevice")

    def hipMalloc(self, nbytes: int) -> int:
        p = ctypes.c_void_p()
        self._check(self.lib.hipMalloc(ctypes.byref(p), nbytes), "hipMalloc")
        return p.value

    def hipMallocFineGrained(self, nbytes: int) -> int:
        r"""System-coherent (fine-grained) allocation — required for buffers
        that peer processes read mid-kernel (hipDeviceMallocFinegrained)."""
        p = ctypes.c_void_p()
        self._check(self.lib.hipExtMallocWithFlags(ctypes.byref(p), nbytes, 0x1),
                    "hipExtMallocWithFlags")
        return p.value

    def hipFree(self, ptr: int):
        self._check(self.lib.hipFree(ctypes.c_void_p(ptr)), "hipFree")

    def hipMemset(self, ptr: int, value: int, nbytes: int):
        self._check(self.lib.hipMemset(ctypes.c_void_p(ptr), value, nbytes),
                    "hipMemset")

    def hipIpcGetMemHandle(self, ptr: int) -> hipIpcMemHandle_t:
        h = hipIpcMemHandle_t()
        self._check(self.lib.hipIpcGetMemHandle(ctypes.byref(h),
                                                ctypes.c_void_p(ptr)),
                    "hipIpcGetMemHandle")
        return h

    def hipIpcOpenMemHandle(self, handle: hipIpcMemHandle_t) -> int:
        p = ctypes.c_void_p()
        # hipIpcMemLazyEnablePeerAccess = 1
        self._check(self.lib.hipIpcOpenMemHandle(ctypes.byref(p), handle, 1),
                    "hipIpcOpenMemHandle")
        return p.value

    def hipIpcCloseMemHandle(self, ptr: int):
        self._check(self.lib.hipIpcCloseMemHandle(ctypes.c_void_p(ptr)),
                    "hipIpcCloseMemHandle")

    def hipDeviceSynchronize(self):
        self._check(self.lib.hipDeviceSynchronize(), "hipDeviceSynchronize")


_rt: Optional[HipRTLibrary] = None


def hip_rt() -> HipRTLibrary:
    global _rt
    if _rt is None:
        _rt = HipRTLibrary()
    return _rt


def create_shared_buffer(nbytes: int, group=None,
                         fine_grained: bool = True) -> List[int]:
    r"""Allocate ``nbytes`` on this rank's GPU, exchange hipIpc handles over
    the (gloo/RCCL) group, and open every peer's buffer: returns one device
    pointer per rank (own rank's is the local allocation). The building
    block for xGMI push-kernel collectives."""
    import torch.distributed as dist

    rt = hip_rt()
    ptr = rt.hipMallocFineGrained(nbytes) if fine_grained else rt.hipMalloc(nbytes)
    handle = rt.hipIpcGetMemHandle(ptr).bytes_()
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    boxes: List[Optional[bytes]] = [None] * world
    dist.all_gather_object(boxes, handle, group=group)
    ptrs = []
    for r in range(world):
        if r == rank:
            ptrs.append(ptr)
        else:
            ptrs.append(rt.hipIpcOpenMemHandle(
                hipIpcMemHandle_t.from_bytes(boxes[r])))
    return ptrs


def free_shared_buffer(ptrs: List[int], group=None) -> None:
    import torch.distributed as dist

    rank = dist.get_rank(group)
    rt = hip_rt()
    for r, p in enumerate(ptrs):
        if p:
            (rt.hipFree if r == rank else rt.hipIpcCloseMemHandle)(p)
