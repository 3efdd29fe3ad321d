"""Declarative logits-processing pipelines (reference parity:
flashinfer/logits_processor/ — LogitsPipe (pipeline.py:33), the processor
set (processors.py: Temperature:90, Softmax:137, TopK:193, TopP:272,
MinP:322, Sample:370), type tagging (types.py) and fusion rules
(fusion_rules.py)). The compiler greedily fuses adjacent processors onto
the sorting-free rejection-sampling kernels (csrc/sampling.hip), e.g.
[Temperature, Softmax] -> one online-softmax launch and
[TopK, TopP, Sample] -> one dual-pivot rejection launch."""
from __future__ import annotations

from enum import Enum
from typing import Callable, List, Optional, Sequence, Tuple

import torch

from . import sampling as _s


class TensorType(Enum):
    LOGITS = "logits"
    PROBS = "probs"
    INDICES = "indices"


class TaggedTensor:
    """A tensor tagged with its pipeline type (reference types.py role)."""

    def __init__(self, data: torch.Tensor, type: TensorType):
        self.data = data
        self.type = type

    @staticmethod
    def logits(data):
        return TaggedTensor(data, TensorType.LOGITS)

    @staticmethod
    def probs(data):
        return TaggedTensor(data, TensorType.PROBS)


class CompileError(ValueError):
    pass


class LegalizationError(ValueError):
    pass


class Op:
    """Typed low-level operator (reference op.py Op:23): consumes and
    produces :class:`TaggedTensor` with declared IN/OUT types."""

    IN: Optional[TensorType] = None
    OUT: Optional[TensorType] = None

    def __init__(self):
        if self.IN is None or self.OUT is None:
            raise ValueError(
                f"Operator {type(self).__name__} must define IN and OUT "
                "tensor types")

    def __call__(self, tensor: TaggedTensor, **kwargs) -> TaggedTensor:
        raise NotImplementedError

    def _validate_input_type(self, tensor: TaggedTensor) -> TensorType:
        if tensor.type != self.IN:
            raise ValueError(
                f"Operator {type(self).__name__} cannot accept input type "
                f"{tensor.type}. Expected: {self.IN}")
        return self.OUT

    def __repr__(self):
        return f"{type(self).__name__}({self.IN} -> {self.OUT})"


class ParameterizedOp(Op):
    """Op with bound default parameters (reference op.py role)."""

    def __init__(self, **params):
        super().__init__()
        self.params = params


class LogitsProcessor:
    """One pipeline stage. ``in_types`` lists accepted input types;
    ``out_type(in_type)`` gives the produced type."""

    params: Tuple[str, ...] = ()

    def __init__(self, **kw):
        self.defaults = kw

    def in_types(self) -> Sequence[TensorType]:
        raise NotImplementedError

    def out_type(self, t: TensorType) -> TensorType:
        raise NotImplementedError

    def apply(self, x: torch.Tensor, t: TensorType, **kw) -> torch.Tensor:
        raise NotImplementedError

    def _get(self, kw, name, required=True):
        if name in kw and kw[name] is not None:
            return kw[name]
        if name in self.defaults:
            return self.defaults[name]
        if required:
            raise ValueError(f"{type(self).__name__} needs runtime param {name!r}")
        return None


class Temperature(LogitsProcessor):
    params = ("temperature",)

    def in_types(self):
        return (TensorType.LOGITS,)

    def out_type(self, t):
        return TensorType.LOGITS

    def apply(self, x, t, **kw):
        temp = self._get(kw, "temperature")
        if torch.is_tensor(temp):
            return x / temp[:, None].to(x.dtype)
        return x / temp


class Softmax(LogitsProcessor):
    def in_types(self):
        return (TensorType.LOGITS,)

    def out_type(self, t):
        return TensorType.PROBS

    def apply(self, x, t, **kw):
        return _s.softmax(x.float(), temperature=1.0)


class TopK(LogitsProcessor):
    r"""LOGITS -> LOGITS (mask to -inf) or PROBS -> PROBS (renormalize)."""

    params = ("top_k",)

    def __init__(self, joint_topk_topp: bool = False, **kw):
        super().__init__(**kw)
        self.joint_topk_topp = joint_topk_topp

    def in_types(self):
        return (TensorType.LOGITS, TensorType.PROBS)

    def out_type(self, t):
        return t

    def apply(self, x, t, **kw):
        k = self._get(kw, "top_k")
        if t == TensorType.LOGITS:
            return _s.top_k_mask_logits(x.float(), k)
        return _s.top_k_renorm_probs(x.float(), k)


class TopP(LogitsProcessor):
    params = ("top_p",)

    def in_types(self):
        return (TensorType.PROBS,)

    def out_type(self, t):
        return TensorType.PROBS

    def apply(self, x, t, **kw):
        return _s.top_p_renorm_probs(x.float(), self._get(kw, "top_p"))


class MinP(LogitsProcessor):
    params = ("min_p",)

    def in_types(self):
        return (TensorType.PROBS,)

    def out_type(self, t):
        return TensorType.PROBS

    def apply(self, x, t, **kw):
        p = self._get(kw, "min_p")
        xf = x.float()
        thr = xf.max(-1, keepdim=True).values
        thr = thr * (p[:, None] if torch.is_tensor(p) else p)
        y = torch.where(xf >= thr, xf, torch.zeros_like(xf))
        return y / y.sum(-1, keepdim=True)


class Sample(LogitsProcessor):
    params = ("generator", "indices")

    def __init__(self, deterministic: bool = True, **kw):
        super().__init__(**kw)
        self.deterministic = deterministic

    def in_types(self):
        return (TensorType.LOGITS, TensorType.PROBS)

    def out_type(self, t):
        return TensorType.INDICES

    def apply(self, x, t, **kw):
        gen = self._get(kw, "generator", required=False)
        ind = self._get(kw, "indices", required=False)
        if t == TensorType.LOGITS:
            return _s.sampling_from_logits(x.float(), ind, generator=gen)
        return _s.sampling_from_probs(x.float(), ind, generator=gen)


class _Fused(LogitsProcessor):
    """A fusion-rule product: one kernel covering several stages."""

    def __init__(self, name, in_t, out_t, fn, params):
        super().__init__()
        self.name = name
        self._in, self._out, self._fn = in_t, out_t, fn
        self.params = params

    def in_types(self):
        return (self._in,)

    def out_type(self, t):
        return self._out

    def apply(self, x, t, **kw):
        return self._fn(x, **kw)


class FusionRule:
    """Matches a window of processors and replaces it with one _Fused op."""

    def __init__(self, pattern: Tuple[type, ...], build: Callable, name: str):
        self.pattern = pattern
        self.build = build
        self.name = name


def _r_temp_softmax(ps):
    t = ps[0]
    return _Fused(
        "temperature_softmax", TensorType.LOGITS, TensorType.PROBS,
        lambda x, **kw: _s.softmax(x.float(),
                                   temperature=t._get(kw, "temperature")),
        ("temperature",))


def _r_topk_topp_sample(ps):
    tk, tp, sm = ps
    return _Fused(
        "top_k_top_p_sample", TensorType.PROBS, TensorType.INDICES,
        lambda x, **kw: _s.top_k_top_p_sampling_from_probs(
            x.float(), tk._get(kw, "top_k"), tp._get(kw, "top_p"),
            sm._get(kw, "indices", required=False),
            generator=sm._get(kw, "generator", required=False)),
        ("top_k", "top_p", "generator", "indices"))


def _r_topk_sample(ps):
    tk, sm = ps
    return _Fused(
        "top_k_sample", TensorType.PROBS, TensorType.INDICES,
        lambda x, **kw: _s.top_k_sampling_from_probs(
            x.float(), tk._get(kw, "top_k"),
            sm._get(kw, "indices", required=False),
            generator=sm._get(kw, "generator", required=False)),
        ("top_k", "generator", "indices"))


def _r_topp_sample(ps):
    tp, sm = ps
    return _Fused(
        "top_p_sample", TensorType.PROBS, TensorType.INDICES,
        lambda x, **kw: _s.top_p_sampling_from_probs(
            x.float(), tp._get(kw, "top_p"),
            sm._get(kw, "indices", required=False),
            generator=sm._get(kw, "generator", required=False)),
        ("top_p", "generator", "indices"))


def _r_minp_sample(ps):
    mp, sm = ps
    return _Fused(
        "min_p_sample", TensorType.PROBS, TensorType.INDICES,
        lambda x, **kw: _s.min_p_sampling_from_probs(
            x.float(), mp._get(kw, "min_p"),
            sm._get(kw, "indices", required=False),
            generator=sm._get(kw, "generator", required=False)),
        ("min_p", "generator", "indices"))


DEFAULT_FUSION_RULES = [
    FusionRule((Temperature, Softmax), _r_temp_softmax, "temp+softmax"),
    FusionRule((TopK, TopP, Sample), _r_topk_topp_sample, "topk+topp+sample"),
    FusionRule((TopK, Sample), _r_topk_sample, "topk+sample"),
    FusionRule((TopP, Sample), _r_topp_sample, "topp+sample"),
    FusionRule((MinP, Sample), _r_minp_sample, "minp+sample"),
]


def legalize_processors(processors, input_type):
    """Type-check the chain; returns the per-stage input types."""
    t = input_type
    chain = []
    for i, p in enumerate(processors):
        if t not in p.in_types():
            raise LegalizationError(
                f"stage {i} ({type(p).__name__}) cannot take {t}")
        chain.append(t)
        t = p.out_type(t)
        if t == TensorType.INDICES and i != len(processors) - 1:
            raise LegalizationError("Sample must be the last stage")
    return chain


def compile_pipeline(processors, input_type,
                     custom_fusion_rules: Optional[List[FusionRule]] = None):
    """Greedy left-to-right pattern fusion. Fusion only fires when the
    window's concrete input type matches the fused kernel's contract
    (e.g. TopK+Sample fuses on PROBS, not on LOGITS)."""
    rules = list(custom_fusion_rules or []) + DEFAULT_FUSION_RULES
    stage_types = legalize_processors(processors, input_type)
    out, i = [], 0
    while i < len(processors):
        fused = None
        for r in rules:
            n = len(r.pattern)
            if i + n <= len(processors) and all(
                type(processors[i + j]) is r.pattern[j] for j in range(n)
            ):
                cand = r.build(processors[i:i + n])
                if stage_types[i] in cand.in_types():
                    fused = (cand, n)
                    break
        if fused:
            out.append(fused[0])
            i += fused[1]
        else:
            out.append(processors[i])
            i += 1
    legalize_processors(out, input_type)
    return out


Compiler = compile_pipeline  # reference-name alias


class LogitsPipe:
    r"""Declarative logits pipeline: ``LogitsPipe([Temperature(), Softmax(),
    TopK(), Sample()])(logits, temperature=0.9, top_k=40)``."""

    def __init__(self, processors: List[LogitsProcessor], compile: bool = True,
                 input_type: Optional[TensorType] = None,
                 custom_fusion_rules: Optional[List[FusionRule]] = None,
                 custom_validity_checks=None):
        if not processors:
            raise ValueError("Pipeline cannot be empty")
        first_in = processors[0].in_types()
        if input_type is None:
            if len(first_in) > 1:
                raise ValueError(
                    "first processor accepts several input types; pass "
                    "input_type=TensorType.LOGITS or .PROBS")
            input_type = first_in[0]
        self.input_type = input_type
        self.processors = list(processors)
        self.compiled_ops: List[LogitsProcessor] = list(processors)
        if compile:
            self.compile(custom_fusion_rules)

    def compile(self, custom_fusion_rules=None):
        self.compiled_ops = compile_pipeline(self.processors, self.input_type,
                                             custom_fusion_rules)
        return self

    def __call__(self, x: torch.Tensor, **params) -> torch.Tensor:
        t = self.input_type
        for op in self.compiled_ops:
            x = op.apply(x, t, **params)
            t = op.out_type(t)
        return x

    run = __call__
