"""Intra-kernel event profiler (reference parity: flashinfer/profiler/
__init__.py decode_tag:34, export_to_perfetto_trace:52). Kernels append
(tag, timestamp) events into a uint64 buffer — see
csrc/include/fi/profiler.hpp for the device side and the buffer layout —
and this module decodes them and exports a chrome://tracing /
Perfetto-loadable JSON (tg4perfetto is not in the image; the JSON format
is the portable equivalent).

Usage::

    buf = fi.profiler.make_profiler_buffer(1 << 16)
    wrapper.run(q, kv, profiler_buffer=buf)
    fi.profiler.export_to_chrome_trace(buf, ["tile", "kv_mainloop"], "t.json")
"""
from __future__ import annotations

import json
from enum import Enum
from typing import List, Tuple

import torch

# s_memrealtime ticks at a constant 100 MHz on CDNA (10 ns / tick)
REALTIME_TICK_NS = 10.0


class EventType(Enum):
    kBegin = 0
    kEnd = 1
    kInstant = 2


def make_profiler_buffer(capacity_slots: int = 1 << 16,
                         device="cuda") -> torch.Tensor:
    r"""Allocate and initialize an event buffer of ``capacity_slots`` uint64
    slots (2 header slots + events)."""
    buf = torch.zeros(capacity_slots, dtype=torch.uint64, device=device)
    buf[0] = capacity_slots
    return buf


def decode_tag(tag: int) -> Tuple[int, int, int]:
    """Split a 32-bit tag into (block_idx, event_idx, event_type)."""
    event_type = tag & 0x3
    event_idx = (tag >> 2) & 0x3FF
    block_idx = (tag >> 12) & 0xFFFFF
    return block_idx, event_idx, event_type


def decode_events(profiler_buffer: torch.Tensor):
    """Return a list of (block_idx, event_idx, event_type, timestamp_ticks),
    with 32-bit timestamp wraparound unrolled per block."""
    host = profiler_buffer.cpu().view(torch.uint64)
    n = min(int(host[1]) + 2, int(host[0]))
    events = []
    for i in range(2, n):
        v = int(host[i])
        if v == 0:
            continue
        tag = v >> 32
        ts = v & 0xFFFFFFFF
        block_idx, event_idx, event_type = decode_tag(tag)
        events.append((block_idx, event_idx, event_type, ts))
    return events


def export_to_chrome_trace(
    profiler_buffer: torch.Tensor,
    event_names: List[str],
    file_name: str,
) -> None:
    r"""Write a chrome://tracing JSON: one "process" per workgroup, duration
    slices per begin/end event pair."""
    events = decode_events(profiler_buffer)
    if not events:
        with open(file_name, "w") as f:
            json.dump({"traceEvents": []}, f)
        return
    t0 = min(ts for _, _, _, ts in events)
    trace = []
    for block_idx, event_idx, event_type, ts in events:
        name = (event_names[event_idx]
                if event_idx < len(event_names) else f"event{event_idx}")
        us = (ts - t0) * REALTIME_TICK_NS / 1000.0
        ph = {0: "B", 1: "E", 2: "i"}[event_type]
        ev = {"name": name, "ph": ph, "ts": us, "pid": 0,
              "tid": block_idx}
        if ph == "i":
            ev["s"] = "t"
        trace.append(ev)
    trace.sort(key=lambda e: e["ts"])
    with open(file_name, "w") as f:
        json.dump({"traceEvents": trace,
                   "displayTimeUnit": "ns"}, f)


# perfetto loads chrome JSON traces directly; keep the reference's name
export_to_perfetto_trace = export_to_chrome_trace
