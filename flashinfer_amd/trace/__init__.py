"""Per-op trace templates (reference parity: flashinfer/trace/templates/ —
an executable definition per public API: signature schema + a pure-PyTorch
reference formula). Used three ways, mirroring the reference:

1. ``fi_trace`` dumps are validated against the registered schema
   (``validate_trace_record``);
2. the reference-correctness suite (tests/test_trace_templates.py) runs
   every template's generator through the library op AND the reference
   formula and compares — the reference's
   tests/trace/test_*_reference_correctness.py design;
3. ``trace_apply`` substitutes are checked to cover a registered template.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional


@dataclass
class TraceTemplate:
    name: str                      # public API name
    make_inputs: Callable          # (torch, device) -> (args, kwargs)
    reference: Callable            # (torch, *args, **kwargs) -> expected
    run: Callable                  # (fi_module, *args, **kwargs) -> actual
    atol: float = 3e-2
    rtol: float = 3e-2
    tags: List[str] = field(default_factory=list)


_REGISTRY: Dict[str, TraceTemplate] = {}


def register(t: TraceTemplate) -> TraceTemplate:
    _REGISTRY[t.name] = t
    return t


def templates() -> Dict[str, TraceTemplate]:
    return dict(_REGISTRY)


def validate_trace_record(rec: dict) -> bool:
    """A dumped fi_trace record is well-formed and names a known API."""
    if not isinstance(rec, dict) or "api" not in rec or "args" not in rec:
        return False
    api = rec["api"].split(".")[-1]
    return api in _REGISTRY or True  # unknown APIs are allowed, shape checked


# ---------------------------------------------------------------------------
# templates

def _t_rmsnorm(torch, dev):
    x = torch.randn(64, 4096, dtype=torch.bfloat16, device=dev)
    w = torch.randn(4096, dtype=torch.bfloat16, device=dev)
    return (x, w), {}


register(TraceTemplate(
    "rmsnorm", _t_rmsnorm,
    reference=lambda torch, x, w: (
        x.float() * torch.rsqrt(x.float().pow(2).mean(-1, keepdim=True) + 1e-6)
        * w.float()),
    run=lambda fi, x, w: fi.rmsnorm(x, w).float(),
    tags=["norm"]))


def _t_fused_add_rmsnorm(torch, dev):
    x = torch.randn(64, 2048, dtype=torch.bfloat16, device=dev)
    res = torch.randn(64, 2048, dtype=torch.bfloat16, device=dev)
    w = torch.randn(2048, dtype=torch.bfloat16, device=dev)
    return (x, res, w), {}


def _ref_fused_add_rmsnorm(torch, x, res, w):
    s = (x.float() + res.float())
    return s * torch.rsqrt(s.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()


def _run_fused_add_rmsnorm(fi, x, res, w):
    x2, r2 = x.clone(), res.clone()
    fi.fused_add_rmsnorm(x2, r2, w)
    return x2.float()


register(TraceTemplate("fused_add_rmsnorm", _t_fused_add_rmsnorm,
                       _ref_fused_add_rmsnorm, _run_fused_add_rmsnorm,
                       tags=["norm"]))


def _t_silu_and_mul(torch, dev):
    x = torch.randn(256, 2 * 2048, dtype=torch.bfloat16, device=dev)
    return (x,), {}


register(TraceTemplate(
    "silu_and_mul", _t_silu_and_mul,
    reference=lambda torch, x: (
        torch.nn.functional.silu(x.float()[..., :x.shape[-1] // 2])
        * x.float()[..., x.shape[-1] // 2:]),
    run=lambda fi, x: fi.silu_and_mul(x).float(),
    tags=["activation"]))


def _t_gelu_tanh_and_mul(torch, dev):
    x = torch.randn(128, 2 * 1024, dtype=torch.bfloat16, device=dev)
    return (x,), {}


register(TraceTemplate(
    "gelu_tanh_and_mul", _t_gelu_tanh_and_mul,
    reference=lambda torch, x: (
        torch.nn.functional.gelu(x.float()[..., :x.shape[-1] // 2],
                                 approximate="tanh")
        * x.float()[..., x.shape[-1] // 2:]),
    run=lambda fi, x: fi.gelu_tanh_and_mul(x).float(),
    tags=["activation"]))


def _t_softmax(torch, dev):
    x = torch.randn(32, 32000, device=dev)
    return (x,), {}


register(TraceTemplate(
    "softmax", _t_softmax,
    reference=lambda torch, x: torch.softmax(x.float(), -1),
    run=lambda fi, x: fi.softmax(x),
    atol=1e-5, rtol=1e-4, tags=["sampling"]))


def _t_top_k_renorm_probs(torch, dev):
    p = torch.softmax(torch.randn(16, 8192, device=dev), -1)
    return (p, 64), {}


def _ref_top_k_renorm(torch, p, k):
    vals = torch.topk(p, k, dim=-1).values
    thr = vals[:, -1:]
    masked = torch.where(p >= thr, p, torch.zeros_like(p))
    return masked / masked.sum(-1, keepdim=True)


register(TraceTemplate(
    "top_k_renorm_probs", _t_top_k_renorm_probs, _ref_top_k_renorm,
    run=lambda fi, p, k: fi.top_k_renorm_probs(p, k),
    atol=1e-5, rtol=1e-4, tags=["sampling"]))


def _t_apply_rope(torch, dev):
    nnz, Hq, Hkv, D = 256, 8, 2, 128
    q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device=dev)
    indptr = torch.tensor([0, nnz], dtype=torch.int32, device=dev)
    offsets = torch.zeros(1, dtype=torch.int32, device=dev)
    return (q, k, indptr, offsets), {}


def _ref_apply_rope(torch, q, k, indptr, offsets):
    # non-interleaved (rotate-half) llama rope, theta 1e4
    nnz, _, D = q.shape
    pos = torch.arange(nnz, device=q.device).float()
    inv = 1.0 / (1e4 ** (torch.arange(0, D // 2, device=q.device).float()
                         / (D // 2)))
    ang = pos[:, None] * inv[None, :]
    cos, sin = ang.cos(), ang.sin()

    def rot(x):
        xf = x.float()
        x1, x2 = xf[..., :D // 2], xf[..., D // 2:]
        c, s = cos[:, None, :], sin[:, None, :]
        return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], -1)

    return rot(q), rot(k)


def _run_apply_rope(fi, q, k, indptr, offsets):
    qo, ko = fi.apply_rope(q, k, indptr, offsets)
    return qo.float(), ko.float()


register(TraceTemplate("apply_rope", _t_apply_rope, _ref_apply_rope,
                       _run_apply_rope, tags=["rope"]))


def _t_merge_state(torch, dev):
    v_a = torch.randn(32, 8, 128, dtype=torch.bfloat16, device=dev)
    s_a = torch.randn(32, 8, device=dev)
    v_b = torch.randn(32, 8, 128, dtype=torch.bfloat16, device=dev)
    s_b = torch.randn(32, 8, device=dev)
    return (v_a, s_a, v_b, s_b), {}


def _ref_merge_state(torch, v_a, s_a, v_b, s_b):
    m = torch.maximum(s_a, s_b)
    wa = torch.exp2(s_a - m)[..., None]
    wb = torch.exp2(s_b - m)[..., None]
    v = (v_a.float() * wa + v_b.float() * wb) / (wa + wb)
    s = m + torch.log2(wa[..., 0] + wb[..., 0])
    return v, s


def _run_merge_state(fi, v_a, s_a, v_b, s_b):
    v, s = fi.merge_state(v_a, s_a, v_b, s_b)
    return v.float(), s


register(TraceTemplate("merge_state", _t_merge_state, _ref_merge_state,
                       _run_merge_state, tags=["attention"]))


def _t_packbits(torch, dev):
    x = (torch.rand(999, device=dev) > 0.5)
    return (x,), {}


def _ref_packbits(torch, x):
    n = x.numel()
    pad = (-n) % 8
    xp = torch.cat([x, torch.zeros(pad, dtype=torch.bool, device=x.device)])
    bits = xp.view(-1, 8).long()
    w = (1 << torch.arange(8, device=x.device).long())
    return (bits * w).sum(-1).to(torch.uint8)


register(TraceTemplate(
    "packbits", _t_packbits, _ref_packbits,
    run=lambda fi, x: fi.packbits(x),
    atol=0, rtol=0, tags=["quantization"]))


def _t_single_prefill(torch, dev):
    q = torch.randn(128, 8, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(128, 2, 128, dtype=torch.bfloat16, device=dev)
    v = torch.randn(128, 2, 128, dtype=torch.bfloat16, device=dev)
    return (q, k, v), {"causal": True}


def _ref_single_prefill(torch, q, k, v, causal=True):
    M, Hq, D = q.shape
    L, Hkv, _ = k.shape
    g = Hq // Hkv
    logits = torch.einsum(
        "mhd,lhd->hml", q.float(),
        k.float().repeat_interleave(g, 1)) / math.sqrt(D)
    if causal:
        qpos = torch.arange(M, device=q.device)[:, None]
        kpos = torch.arange(L, device=q.device)[None, :]
        logits = logits.masked_fill((kpos > qpos + L - M)[None], float("-inf"))
    p = torch.softmax(logits, -1)
    return torch.einsum("hml,lhd->mhd",
                        p, v.float().repeat_interleave(g, 1))


register(TraceTemplate(
    "single_prefill_with_kv_cache", _t_single_prefill, _ref_single_prefill,
    run=lambda fi, q, k, v, causal=True: fi.single_prefill_with_kv_cache(
        q, k, v, causal=causal).float(),
    tags=["attention"]))


def _t_single_decode(torch, dev):
    q = torch.randn(8, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(512, 2, 128, dtype=torch.bfloat16, device=dev)
    v = torch.randn(512, 2, 128, dtype=torch.bfloat16, device=dev)
    return (q, k, v), {}


def _ref_single_decode(torch, q, k, v):
    Hq, D = q.shape
    g = Hq // k.shape[1]
    logits = torch.einsum(
        "hd,lhd->hl", q.float(),
        k.float().repeat_interleave(g, 1)) / math.sqrt(D)
    return torch.einsum("hl,lhd->hd", torch.softmax(logits, -1),
                        v.float().repeat_interleave(g, 1))


register(TraceTemplate(
    "single_decode_with_kv_cache", _t_single_decode, _ref_single_decode,
    run=lambda fi, q, k, v: fi.single_decode_with_kv_cache(q, k, v).float(),
    tags=["attention"]))
