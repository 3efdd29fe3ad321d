"""Decode attention: single-request API + batch plan/run wrappers.

Reference parity: flashinfer/decode.py (single_decode_with_kv_cache:515,
BatchDecodeWithPagedKVCacheWrapper:751 with plan:1280/run:1851,
CUDAGraphBatchDecodeWithPagedKVCacheWrapper:2334).

MI355X design: one hand-written CDNA4 kernel (csrc/attention/batch_decode.hip)
covers single and batch decode over paged KV. The CPU planner splits each
request's KV into chunks sized so the launch fills the 256-CU chip
(grid = n_items x num_kv_heads 1-wave blocks), partials always go through the
LSE merge kernel (split-KV merge, reference scheduler.cuh:349 role), and the
run path is hipGraph-capturable (fixed grids, no allocation, no H2D).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple, Union

import torch

from ._lib import get_ext
from .api_logging import flashinfer_api
from .fi_trace import fi_trace
from .utils import (
    WorkspaceAllocator,
    default_sm_scale,
    layout_code,
    unpack_paged_kv_cache,
)

_TARGET_BLOCKS = 8192  # 1-wave blocks: ~32 per CU keeps enough loads in flight

# Whole-request fused kernel (one workgroup per (req, kv_head), in-LDS wave
# merge, single launch): wins while the per-wave KV slice stays latency-sized;
# past this the split path's finer-grain work items win back. Measured
# crossover on MI355X (profiles/README r02 fused-decode entry).
_FUSED_MAX_KV = 2048

# MFMA fused decode (GQA group >= 8): one workgroup per (req, kv_head), the
# whole group as the 32-col q dimension of 32x32 MFMA tiles, 4 barrier-free
# waves splitting the KV, in-LDS merge. The small-batch/short-kv GQA shape
# where both the vector kernel (8 dot+shfl chains per K read) and the
# prefill-reuse path (94%-padded 128-row tiles) lose.
_MFMA_MAX_KV = 4096
# fp8 e4m3 KV has no tensor-core fallback (the prefill-based tc path needs
# kv_dt == q_dt) and the split-vector fp8 dequant is scalar (~0.39 TB/s at
# bs=128/kv=4096 vs 2.2 TB/s on the MFMA route — profiles/README.md r02
# addendum), so for fp8 KV the MFMA route applies at any length: its per-wave
# KV slices + cross-WG split scale with kv_len with no structural cap.
_MFMA_MAX_KV_F8 = 1 << 30


def _plan_chunks(kv_lens, num_kv_heads: int, page_size: int,
                 fixed_split_size=None, disable_split_kv=False):
    """Pick a split-KV chunk size and emit work items.

    fixed_split_size / disable_split_kv give batch-invariant deterministic
    partitioning (reference decode.py:1354 plan args).
    Returns (chunk_size, work_req, work_chunk, merge_indptr) as Python lists.
    """
    batch = len(kv_lens)
    total = sum(kv_lens)
    max_len = max(kv_lens) if batch else 0
    if disable_split_kv:
        chunk = max(max_len, page_size)
    elif fixed_split_size is not None:
        chunk = int(fixed_split_size)
    else:
        items_target = max(batch, _TARGET_BLOCKS // max(1, num_kv_heads))
        # no fixed token floor: at small batch*kv the wave count is the
        # bottleneck (1 wave/SIMD leaves the VALU/DS latency chains exposed —
        # measured 139us -> 30us at bs=16/kv=1024/GQA-8 by splitting finer)
        chunk = max(64, page_size, math.ceil(total / max(1, items_target)))
    # round up to page multiple so chunks don't straddle partially-read pages
    chunk = (chunk + page_size - 1) // page_size * page_size
    work_req, work_chunk, merge_indptr = [], [], [0]
    for b, L in enumerate(kv_lens):
        n = max(1, math.ceil(L / chunk))
        for c in range(n):
            work_req.append(b)
            work_chunk.append(c)
        merge_indptr.append(merge_indptr[-1] + n)
    return chunk, work_req, work_chunk, merge_indptr


class BatchDecodeWithPagedKVCacheWrapper:
    r"""Batch decode attention over a paged KV cache (plan/run API)."""

    def __init__(
        self,
        float_workspace_buffer: torch.Tensor,
        kv_layout: str = "NHD",
        use_cuda_graph: bool = False,
        use_tensor_cores: Optional[bool] = None,
        paged_kv_indptr_buffer: Optional[torch.Tensor] = None,
        paged_kv_indices_buffer: Optional[torch.Tensor] = None,
        paged_kv_last_page_len_buffer: Optional[torch.Tensor] = None,
        backend: str = "fa2",
        jit_args=None,
    ) -> None:
        self._float_workspace_buffer = float_workspace_buffer
        self.device = float_workspace_buffer.device
        self._int_workspace_buffer = torch.empty(
            8 * 1024 * 1024, dtype=torch.uint8, device=self.device
        )
        self._kv_layout = kv_layout
        self._use_cuda_graph = use_cuda_graph
        self._use_tensor_cores = use_tensor_cores
        self._fixed_indptr_buf = paged_kv_indptr_buffer
        self._fixed_indices_buf = paged_kv_indices_buffer
        self._fixed_last_page_len_buf = paged_kv_last_page_len_buffer
        self._plan_info = None

    @property
    def is_cuda_graph_enabled(self) -> bool:
        return self._use_cuda_graph

    def reset_workspace_buffer(
        self, float_workspace_buffer: torch.Tensor, int_workspace_buffer: torch.Tensor
    ) -> None:
        self._float_workspace_buffer = float_workspace_buffer
        self._int_workspace_buffer = int_workspace_buffer

    def plan(
        self,
        indptr: torch.Tensor,
        indices: torch.Tensor,
        last_page_len: torch.Tensor,
        num_qo_heads: int,
        num_kv_heads: int,
        head_dim: int,
        page_size: int,
        pos_encoding_mode: str = "NONE",
        window_left: int = -1,
        logits_soft_cap: Optional[float] = None,
        q_data_type: Optional[torch.dtype] = torch.bfloat16,
        kv_data_type: Optional[torch.dtype] = None,
        data_type: Optional[torch.dtype] = None,
        sm_scale: Optional[float] = None,
        rope_scale: Optional[float] = None,
        rope_theta: Optional[float] = None,
        non_blocking: bool = True,
        fixed_split_size: Optional[int] = None,
        disable_split_kv: bool = False,
        **kwargs,
    ) -> None:
        if pos_encoding_mode not in ("NONE", "ALIBI"):
            raise NotImplementedError(
                "pos_encoding_mode must be NONE or ALIBI; apply RoPE beforehand"
            )
        # Kernel routing (3 decode shapes, auto-picked unless the caller pins
        # one — use_tensor_cores: None=auto, True=force prefill-MFMA path,
        # False=never (advisor r01: explicit False must be respected)):
        #   1. fused whole-request kernel — short KV: one workgroup per
        #      (req, kv_head), in-LDS merge, single launch (latency shape);
        #   2. tensor-core (prefill-MFMA) path — GQA group >= 8 at long KV:
        #      the whole group costs one 32x32 MFMA tile row instead of 8
        #      VALU dot chains (reference decode.py:1697 design, measured
        #      1.4-1.9x on MI355X, profiles/README r01);
        #   3. split vector kernel + LSE merge — everything else.
        group = num_qo_heads // max(1, num_kv_heads)
        kv_dt = kv_data_type or q_data_type
        batch = indptr.shape[0] - 1
        indptr_h = indptr.to("cpu", torch.int64)
        lp_h = last_page_len.to("cpu", torch.int64)
        np_ = indptr_h[1:] - indptr_h[:-1]
        kv_lens = (
            torch.clamp(np_ - 1, min=0) * page_size
            + torch.where(np_ > 0, lp_h, torch.zeros_like(lp_h))
        ).tolist()
        max_len = max(kv_lens) if kv_lens else 0

        fused_ok = (
            head_dim in (64, 128, 256)
            and group in (1, 2, 4)  # GQA >= 8 goes to the MFMA decode kernel
            and max_len <= _FUSED_MAX_KV
            and fixed_split_size is None
            and self._use_tensor_cores is not True
        )
        # any group <= 32 rides the 32x32 MFMA tile (q dim zero-padded, so
        # per-KV-tile cost is group-independent); for bf16/f16 the fused
        # vector kernel keeps precedence where it is eligible (group <= 4,
        # short kv), for fp8 KV the MFMA route always wins (the vector
        # kernels' fp8 dequant path is scalar — 5.6x slower measured).
        mfma_ok = (
            head_dim in (64, 128)
            and group in (1, 2, 4, 5, 6, 7, 8, 16, 32)
            and q_data_type in (torch.bfloat16, torch.float16)
            and kv_dt in (q_data_type, torch.float8_e4m3fn)
            # bf16/f16 precedence vs the fused vector kernel (measured,
            # profiles/README r02 addendum 6): GROUP >= 4 prefers MFMA up
            # to ~384 (req, kv_head) units (20.9 vs 31.1 us at bs=16/
            # kv=1024 G=4; crossover between 384 and 512 units where the
            # 8-wave WGs exceed one dispatch round); GROUP 1-2 stay on the
            # fused kernel. fp8 KV always prefers MFMA.
            and (max_len <= _MFMA_MAX_KV_F8 if kv_dt != q_data_type
                 else (max_len <= _MFMA_MAX_KV
                       and (not fused_ok
                            or (group >= 4
                                and batch * num_kv_heads <= 384))))
            and fixed_split_size is None
            # use_tensor_cores=True with an fp8 KV cache means THIS kernel:
            # the prefill-based tc path needs kv_dt == q_dt, and the MFMA
            # decode kernel is the matrix-core path for fp8
            and (self._use_tensor_cores is not True or kv_dt != q_data_type)
            and self._use_tensor_cores is not False
        )
        # the split-vector kernel instantiates groups {1,2,4,5,6,7,8,16}
        # only; any other group (3, 12, 24, 64, ...) must take the
        # group-agnostic prefill-based tensor-core path rather than fail
        # at dispatch
        vector_group_ok = group in (1, 2, 4, 5, 6, 7, 8, 16)
        self._tc = (
            not fused_ok and not mfma_ok
            and kv_dt == q_data_type
            and (
                self._use_tensor_cores is True
                or (self._use_tensor_cores is None
                    and (group >= 8 or not vector_group_ok)
                    and fixed_split_size is None and not disable_split_kv)
                or not vector_group_ok
            )
        )
        self._fused = (fused_ok or mfma_ok) and not self._tc
        self._fused_mfma = mfma_ok and not self._tc
        # autotuner hook (reference tactic system role): in fully-auto mode
        # the route choice is a tactic keyed on the bucketed shape — a cached
        # winner overrides the heuristic; under autotune() the first run()
        # profiles every eligible route on the real inputs (see run()).
        self._tune = None
        if (self._use_tensor_cores is None and fixed_split_size is None
                and not disable_split_kv):
            from . import autotuner as _at

            # candidates reflect STRUCTURAL capability (what each kernel
            # family can run), not the heuristic's preference — autotune()
            # exists to discover the crossovers the heuristic approximates
            mfma_possible = (
                head_dim in (64, 128)
                and group in (1, 2, 4, 5, 6, 7, 8, 16, 32)
                and q_data_type in (torch.bfloat16, torch.float16)
                and kv_dt in (q_data_type, torch.float8_e4m3fn))
            routes = []
            if fused_ok:
                routes.append("fused")
            if mfma_possible:
                routes.append("mfma")
            if kv_dt == q_data_type:
                routes.append("tc")
            if vector_group_ok:
                routes.append("split")
            key = (f"decode_route:({_at._bucket(max(1, batch))},"
                   f"{_at._bucket(max(1, max_len))},{group},{head_dim})")
            cached = _at._cache.get(key)
            if cached is not None and routes:
                route = routes[cached % len(routes)]
                self._fused = route in ("fused", "mfma")
                self._fused_mfma = route == "mfma"
                self._tc = route == "tc"
            elif _at._tuning_enabled and len(routes) > 1:
                self._tune = (key, routes,
                              (indptr, indices, last_page_len, num_qo_heads,
                               num_kv_heads, head_dim, page_size,
                               pos_encoding_mode, window_left,
                               logits_soft_cap, q_data_type, kv_data_type,
                               sm_scale))
        if self._tc:
            from .prefill import BatchPrefillWithPagedKVCacheWrapper

            if getattr(self, "_tc_wrapper", None) is None:
                self._tc_wrapper = BatchPrefillWithPagedKVCacheWrapper(
                    self._float_workspace_buffer, self._kv_layout)
            batch = indptr.shape[0] - 1
            qo_indptr = torch.arange(0, batch + 1, dtype=torch.int32)
            self._tc_wrapper.plan(
                qo_indptr, indptr, indices, last_page_len, num_qo_heads,
                num_kv_heads, head_dim, page_size, causal=False,
                pos_encoding_mode=pos_encoding_mode, sm_scale=sm_scale,
                window_left=window_left, logits_soft_cap=logits_soft_cap,
                q_data_type=q_data_type, kv_data_type=kv_data_type,
                non_blocking=non_blocking)
            self._plan_info = dict(
                batch=batch, num_qo_heads=num_qo_heads,
                sm_scale=sm_scale if sm_scale is not None
                else default_sm_scale(head_dim))
            return

        if self._fused:
            dev = self.device

            def _to_dev_f(x, buf):
                xt = x.to(dev, torch.int32, non_blocking=non_blocking)
                if buf is not None:
                    buf[: xt.numel()].copy_(xt)
                    return buf[: xt.numel()]
                return xt

            self._indptr_d = _to_dev_f(indptr, self._fixed_indptr_buf)
            self._indices_d = _to_dev_f(indices, self._fixed_indices_buf)
            self._last_page_len_d = _to_dev_f(
                last_page_len, self._fixed_last_page_len_buf)
            # cross-WG split (mfma path): fill the 256-CU chip when
            # batch x kv_heads alone cannot. Target ONE WG per CU: the
            # kernel is 8 waves at 256 VGPRs (2 waves/SIMD), so a CU hosts
            # exactly one WG — more than 256 WGs means a second dispatch
            # round. Measured at bs=16/kv=1024 GQA-8 (128 units): split=2
            # (256 WGs, 1 round) 22.1 us; split=4 (512 WGs, 2 rounds)
            # 27.7 us; split=8 (4 rounds) 43 us — profiles/README r02.
            self._mfma_split = 1
            if self._fused_mfma and not disable_split_kv:
                units = batch * num_kv_heads
                want = min(max(1, 256 // units), max(1, max_len // 256), 8)
                if want > 1:
                    self._mfma_split = int(want)
                    alloc = WorkspaceAllocator(self._float_workspace_buffer)
                    ni = batch * self._mfma_split
                    self._mfma_tmp_v = alloc.alloc(
                        ni * num_qo_heads * head_dim * 4, torch.float32,
                        (ni, num_qo_heads, head_dim))
                    self._mfma_tmp_s = alloc.alloc(
                        ni * num_qo_heads * 4, torch.float32,
                        (ni, num_qo_heads))
                    self._mfma_merge_indptr = torch.arange(
                        0, (batch + 1) * self._mfma_split, self._mfma_split,
                        dtype=torch.int32).to(dev, non_blocking=non_blocking)
                    # same-XCD in-kernel merge: the dispatcher round-robins
                    # linear workgroup ids across the 8 XCDs, so when
                    # batch*kv_heads % 8 == 0 every z-WG of a (req, kv_head)
                    # lands on one XCD and the last arriver merges in-kernel
                    # (no merge_states launch). Counters self-reset; zeroed
                    # once here (hipGraph-safe).
                    self._mfma_counters = None
                    if units % 8 == 0:
                        self._mfma_counters = torch.zeros(
                            units, dtype=torch.int32, device=dev)
            self._plan_info = dict(
                batch=batch, num_qo_heads=num_qo_heads,
                num_kv_heads=num_kv_heads, head_dim=head_dim,
                page_size=page_size, window_left=window_left,
                logits_soft_cap=float(logits_soft_cap or 0.0),
                sm_scale=sm_scale if sm_scale is not None
                else default_sm_scale(head_dim),
                q_data_type=q_data_type, alibi=pos_encoding_mode == "ALIBI",
            )
            return

        chunk, work_req, work_chunk, merge_indptr = _plan_chunks(
            kv_lens, num_kv_heads, page_size, fixed_split_size, disable_split_kv
        )
        n_items = len(work_req)

        dev = self.device
        # device-resident page table (reuse fixed buffers under CUDA graph)
        def _to_dev(x, buf):
            xt = x.to(dev, torch.int32, non_blocking=non_blocking)
            if buf is not None:
                buf[: xt.numel()].copy_(xt)
                return buf[: xt.numel()]
            return xt

        self._indptr_d = _to_dev(indptr, self._fixed_indptr_buf)
        self._indices_d = _to_dev(indices, self._fixed_indices_buf)
        self._last_page_len_d = _to_dev(last_page_len, self._fixed_last_page_len_buf)

        meta = torch.tensor(
            work_req + work_chunk + merge_indptr, dtype=torch.int32
        ).to(dev, non_blocking=non_blocking)
        self._work_req_d = meta[:n_items]
        self._work_chunk_d = meta[n_items : 2 * n_items]
        self._merge_indptr_d = meta[2 * n_items :]

        alloc = WorkspaceAllocator(self._float_workspace_buffer)
        self._tmp_v = alloc.alloc(
            n_items * num_qo_heads * head_dim * 4, torch.float32,
            (n_items, num_qo_heads, head_dim),
        )
        self._tmp_s = alloc.alloc(
            n_items * num_qo_heads * 4, torch.float32, (n_items, num_qo_heads)
        )

        self._plan_info = dict(
            batch=batch, n_items=n_items, chunk=chunk,
            num_qo_heads=num_qo_heads, num_kv_heads=num_kv_heads,
            head_dim=head_dim, page_size=page_size,
            window_left=window_left,
            logits_soft_cap=float(logits_soft_cap or 0.0),
            sm_scale=sm_scale if sm_scale is not None else default_sm_scale(head_dim),
            q_data_type=q_data_type, alibi=pos_encoding_mode == "ALIBI",
        )

    begin_forward = plan

    def run(
        self,
        q: torch.Tensor,
        paged_kv_cache: Union[torch.Tensor, Tuple[torch.Tensor, torch.Tensor]],
        q_scale: Optional[float] = None,
        k_scale: Optional[float] = None,
        v_scale: Optional[float] = None,
        out: Optional[torch.Tensor] = None,
        lse: Optional[torch.Tensor] = None,
        return_lse: bool = False,
        sinks: Optional[torch.Tensor] = None,
        **kwargs,
    ):
        pi = self._plan_info
        if pi is None:
            raise RuntimeError("must call plan() before run()")
        if q.dim() == 2:
            q = q.unsqueeze(1)  # [B, D] MQA convenience
        if getattr(self, "_tune", None) is not None:
            self._profile_routes(q, paged_kv_cache)
        if getattr(self, "_tc", False):
            need_lse = return_lse or sinks is not None
            # q_scale and k_scale both act on the logits: fold into the
            # inner wrapper's k_scale (it multiplies sm_scale at run time)
            ks = None
            if q_scale is not None or k_scale is not None:
                ks = (q_scale or 1.0) * (k_scale or 1.0)
            res = self._tc_wrapper.run(
                q, paged_kv_cache, k_scale=ks, v_scale=v_scale, out=out,
                lse=lse, return_lse=need_lse)
            out, lse = res if need_lse else (res, None)
            if sinks is not None:
                import math as _m

                w = 1.0 / (1.0 + torch.exp2(
                    sinks.float()[None, :] * _m.log2(_m.e) - lse))
                out = (out.float() * w[..., None]).to(out.dtype)
            return (out, lse) if return_lse else out
        k_cache, v_cache = unpack_paged_kv_cache(paged_kv_cache, self._kv_layout)
        sm_scale = pi["sm_scale"]
        if q_scale is not None:
            sm_scale *= q_scale
        if k_scale is not None:
            sm_scale *= k_scale
        if getattr(self, "_fused", False):
            if out is None:
                out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
            need_lse = return_lse or sinks is not None
            if need_lse and lse is None:
                lse = torch.empty((pi["batch"], pi["num_qo_heads"]),
                                  dtype=torch.float32, device=q.device)
            if getattr(self, "_fused_mfma", False):
                sp = getattr(self, "_mfma_split", 1)
                get_ext().batch_decode_mfma_run(
                    q, k_cache, v_cache,
                    self._indices_d, self._indptr_d, self._last_page_len_d,
                    layout_code(self._kv_layout), out,
                    lse if need_lse else None,
                    sm_scale, pi["logits_soft_cap"], pi["window_left"],
                    pi["alibi"], sp,
                    self._mfma_tmp_v if sp > 1 else None,
                    self._mfma_tmp_s if sp > 1 else None,
                    getattr(self, "_mfma_counters", None) if sp > 1 else None,
                )
                if sp > 1 and getattr(self, "_mfma_counters", None) is None:
                    get_ext().merge_states(
                        self._mfma_tmp_v, self._mfma_tmp_s, out, lse,
                        self._mfma_merge_indptr, 0, pi["batch"])
            else:
                get_ext().batch_decode_fused_run(
                    q, k_cache, v_cache,
                    self._indices_d, self._indptr_d, self._last_page_len_d,
                    layout_code(self._kv_layout), out,
                    lse if need_lse else None,
                    sm_scale, pi["logits_soft_cap"], pi["window_left"],
                    pi["alibi"],
                )
            if sinks is not None:
                import math as _m

                w = 1.0 / (1.0 + torch.exp2(
                    sinks.float()[None, :] * _m.log2(_m.e) - lse))
                out = (out.float() * w[..., None]).to(out.dtype)
            if v_scale is not None:
                out = out * v_scale
            return (out, lse) if return_lse else out
        get_ext().batch_decode_run(
            q, k_cache, v_cache,
            self._indices_d, self._indptr_d, self._last_page_len_d,
            layout_code(self._kv_layout),
            self._work_req_d, self._work_chunk_d, pi["chunk"],
            self._tmp_v, self._tmp_s,
            sm_scale, pi["logits_soft_cap"], pi["window_left"], pi["alibi"],
        )
        if out is None:
            out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        if (return_lse or sinks is not None) and lse is None:
            lse = torch.empty(
                (pi["batch"], pi["num_qo_heads"]), dtype=torch.float32, device=q.device
            )
        get_ext().merge_states(
            self._tmp_v, self._tmp_s, out, lse, self._merge_indptr_d, 0, pi["batch"]
        )
        if sinks is not None:
            # attention sink: a per-head virtual logit enters the softmax
            # denominator only: out *= 1 / (1 + e^sink / 2^lse)
            import math as _m

            w = 1.0 / (1.0 + torch.exp2(
                sinks.float()[None, :] * _m.log2(_m.e) - lse))
            out = (out.float() * w[..., None]).to(out.dtype)
        if v_scale is not None:
            out = out * v_scale
        return (out, lse) if return_lse else out

    forward = run

    def _profile_routes(self, q, paged_kv_cache):
        """First run under autotune(): time every eligible route on the real
        inputs, cache the winner under the bucketed shape key, and adopt it
        (reference autotuner.py:98-190 tactic-profiling role)."""
        import time as _time

        import torch as _t

        from . import autotuner as _at
        from . import decode as _dec

        key, routes, plan_args = self._tune
        self._tune = None
        (indptr, indices, last_page_len, Hq, Hkv, D, page, pem, wl, cap,
         qdt, kdt, sm) = plan_args

        def make(route):
            saved = (_dec._FUSED_MAX_KV, _dec._MFMA_MAX_KV,
                     _dec._MFMA_MAX_KV_F8)
            utc = None
            try:
                if route == "split":
                    _dec._FUSED_MAX_KV = _dec._MFMA_MAX_KV = 0
                    _dec._MFMA_MAX_KV_F8 = 0
                    utc = False
                elif route == "tc":
                    utc = True
                elif route == "mfma":
                    _dec._FUSED_MAX_KV = 0
                    _dec._MFMA_MAX_KV = 1 << 30
                else:  # fused (also pin mfma off: the mfma gate yields to
                    # fused only below 384 units, and forcing fused must
                    # actually run the fused kernel)
                    _dec._FUSED_MAX_KV = 1 << 30
                    _dec._MFMA_MAX_KV = 0
                    _dec._MFMA_MAX_KV_F8 = 0
                w = BatchDecodeWithPagedKVCacheWrapper(
                    self._float_workspace_buffer, self._kv_layout,
                    use_tensor_cores=utc)
                prev = _at._tuning_enabled
                _at._tuning_enabled = False
                _at._cache.pop(key, None)
                try:
                    w.plan(indptr, indices, last_page_len, Hq, Hkv, D, page,
                           pos_encoding_mode=pem, window_left=wl,
                           logits_soft_cap=cap, q_data_type=qdt,
                           kv_data_type=kdt, sm_scale=sm)
                finally:
                    _at._tuning_enabled = prev
            finally:
                (_dec._FUSED_MAX_KV, _dec._MFMA_MAX_KV,
                 _dec._MFMA_MAX_KV_F8) = saved
            return w

        timings = []
        for route in routes:
            try:
                w = make(route)
                w.run(q, paged_kv_cache)
                _t.cuda.synchronize()
                t0 = _time.perf_counter()
                for _ in range(3):
                    w.run(q, paged_kv_cache)
                _t.cuda.synchronize()
                timings.append((_time.perf_counter() - t0, route))
            except Exception:
                continue
        if timings:
            best = min(timings)[1]
            _at._cache[key] = routes.index(best)
            mine = ("tc" if self._tc else
                    "mfma" if getattr(self, "_fused_mfma", False) else
                    "fused" if self._fused else "split")
            if best != mine:
                self.plan(indptr, indices, last_page_len, Hq, Hkv, D, page,
                          pos_encoding_mode=pem, window_left=wl,
                          logits_soft_cap=cap, q_data_type=qdt,
                          kv_data_type=kdt, sm_scale=sm)

    def end_forward(self) -> None:
        pass


class CUDAGraphBatchDecodeWithPagedKVCacheWrapper(BatchDecodeWithPagedKVCacheWrapper):
    r"""CUDA-graph (hipGraph) friendly variant: fixed buffers, fixed grids."""

    def __init__(
        self,
        workspace_buffer: torch.Tensor,
        indptr_buffer: torch.Tensor,
        indices_buffer: torch.Tensor,
        last_page_len_buffer: torch.Tensor,
        kv_layout: str = "NHD",
        use_tensor_cores: bool = False,
    ) -> None:
        super().__init__(
            workspace_buffer, kv_layout, use_cuda_graph=True,
            use_tensor_cores=use_tensor_cores,
            paged_kv_indptr_buffer=indptr_buffer,
            paged_kv_indices_buffer=indices_buffer,
            paged_kv_last_page_len_buffer=last_page_len_buffer,
        )


@flashinfer_api
@fi_trace
def single_decode_with_kv_cache(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    kv_layout: str = "NHD",
    pos_encoding_mode: str = "NONE",
    use_tensor_cores: bool = False,
    q_scale: Optional[float] = None,
    k_scale: Optional[float] = None,
    v_scale: Optional[float] = None,
    window_left: int = -1,
    logits_soft_cap: Optional[float] = None,
    sm_scale: Optional[float] = None,
    rope_scale: Optional[float] = None,
    rope_theta: Optional[float] = None,
    return_lse: bool = False,
):
    r"""Decode attention for a single request with contiguous KV.

    q: [num_qo_heads, head_dim]; k/v: [kv_len, num_kv_heads, head_dim] (NHD)
    or [num_kv_heads, kv_len, head_dim] (HND).
    """
    if pos_encoding_mode not in ("NONE", "ALIBI"):
        raise NotImplementedError("apply RoPE beforehand")
    head_dim = q.shape[-1]
    num_qo_heads = q.shape[0]
    if kv_layout == "NHD":
        kv_len, num_kv_heads = k.shape[0], k.shape[1]
    else:
        num_kv_heads, kv_len = k.shape[0], k.shape[1]
    dev = q.device
    if sm_scale is None:
        sm_scale = default_sm_scale(head_dim)
    if q_scale is not None:
        sm_scale *= q_scale
    if k_scale is not None:
        sm_scale *= k_scale

    # view contiguous KV as one giant page
    k4 = k.unsqueeze(0)
    v4 = v.unsqueeze(0)
    items = max(1, min(32, _TARGET_BLOCKS // max(1, num_kv_heads)))
    chunk = max(256, math.ceil(kv_len / items))
    n_items = max(1, math.ceil(kv_len / chunk))
    meta = torch.tensor(
        [0] * n_items + list(range(n_items)) + [0, n_items], dtype=torch.int32
    ).to(dev)
    work_req = meta[:n_items]
    work_chunk = meta[n_items : 2 * n_items]
    merge_indptr = meta[2 * n_items :]
    page_meta = torch.tensor([0, 0, 1, kv_len], dtype=torch.int32).to(dev)
    indices, indptr, last_page_len = page_meta[:1], page_meta[1:3], page_meta[3:]
    tmp_v = torch.empty((n_items, num_qo_heads, head_dim), dtype=torch.float32, device=dev)
    tmp_s = torch.empty((n_items, num_qo_heads), dtype=torch.float32, device=dev)
    get_ext().batch_decode_run(
        q.unsqueeze(0), k4, v4, indices, indptr, last_page_len,
        layout_code(kv_layout), work_req, work_chunk, chunk, tmp_v, tmp_s,
        sm_scale, float(logits_soft_cap or 0.0), window_left,
        pos_encoding_mode == "ALIBI",
    )
    out = torch.empty_like(q.unsqueeze(0))
    lse = (
        torch.empty((1, num_qo_heads), dtype=torch.float32, device=dev)
        if return_lse
        else None
    )
    get_ext().merge_states(tmp_v, tmp_s, out, lse, merge_indptr, 0, 1)
    out = out.squeeze(0)
    if v_scale is not None:
        out = out * v_scale
    return (out, lse.squeeze(0)) if return_lse else out


def fast_decode_plan(wrapper, *args, **kwargs):
    r"""SGLang-style replanning hook (reference decode.py:3826): same contract
    as :meth:`BatchDecodeWithPagedKVCacheWrapper.plan`, kept as a free
    function so serving engines can monkey-patch it."""
    return BatchDecodeWithPagedKVCacheWrapper.plan(wrapper, *args, **kwargs)
