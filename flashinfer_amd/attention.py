"""Holistic mixed-batch attention (reference parity: flashinfer/attention/
_core.py BatchAttention:44, persistent kernel include/flashinfer/attention/
persistent.cuh + TwoStageHolisticPlan scheduler.cuh:1243).

MI355X design (csrc/attention/batch_attention.hip): ONE persistent kernel
(a workgroup per CU) drains an atomic work queue of tagged items — 256-row
prefill MFMA tiles and 32-row MFMA decode items (the GQA group as the q
dimension, in-item LDS merge, direct output: no reduction stage). The host
orders items most-expensive-first; the ticket gives dynamic load balance.
Falls back to the batch-prefill wrapper for shapes outside the persistent
kernel's coverage (hd 192/256, fp8 KV, custom masks, GQA group != 8 decode
rows become prefill tiles)."""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from .prefill import BatchPrefillWithPagedKVCacheWrapper


class BatchAttention:
    def __init__(self, kv_layout: str = "NHD", device=None, **kwargs):
        self._kv_layout = kv_layout
        self._wrapper: Optional[BatchPrefillWithPagedKVCacheWrapper] = None
        self._persistent = False

    def plan(
        self,
        qo_indptr: torch.Tensor,
        kv_indptr: torch.Tensor,
        kv_indices: torch.Tensor,
        kv_len_arr: torch.Tensor,
        num_qo_heads: int,
        num_kv_heads: int,
        head_dim_qk: int,
        head_dim_vo: int,
        page_size: int,
        causal: bool = True,
        sm_scale: Optional[float] = None,
        logits_soft_cap: Optional[float] = None,
        q_data_type: torch.dtype = torch.bfloat16,
        kv_data_type: Optional[torch.dtype] = None,
        use_profiler: bool = False,
        **kwargs,
    ) -> None:
        from .utils import default_sm_scale

        dev = qo_indptr.device if qo_indptr.is_cuda else (
            kv_indices.device if kv_indices.is_cuda else torch.device("cuda"))
        group = num_qo_heads // max(1, num_kv_heads)
        kv_dt = kv_data_type or q_data_type
        self._persistent = (
            head_dim_qk == head_dim_vo
            and head_dim_qk in (64, 128)
            and q_data_type in (torch.bfloat16, torch.float16)
            and kv_dt == q_data_type
        )
        if not self._persistent:
            if self._wrapper is None:
                ws = torch.empty(128 << 20, dtype=torch.uint8, device=dev)
                self._wrapper = BatchPrefillWithPagedKVCacheWrapper(
                    ws, self._kv_layout)
            last_page_len = ((kv_len_arr.to(torch.int64) - 1) % page_size + 1
                             ).to(torch.int32)
            self._wrapper.plan(
                qo_indptr, kv_indptr, kv_indices, last_page_len,
                num_qo_heads, num_kv_heads, head_dim_qk, page_size,
                causal=causal, sm_scale=sm_scale,
                logits_soft_cap=logits_soft_cap, q_data_type=q_data_type)
            return
        qi = qo_indptr.to("cpu", torch.int64)
        qo_lens = (qi[1:] - qi[:-1]).tolist()
        kv_lens = kv_len_arr.to("cpu", torch.int64).tolist()
        self._group_dec = 8 if group == 8 else 0
        # tagged work items: prefill tiles then decode items, each pool
        # most-expensive-first; the decode WG pool is sized by cost fraction
        pf_items, dec_items = [], []  # (cost, kind, req, a, b)
        for r, (ql, kl) in enumerate(zip(qo_lens, kv_lens)):
            if ql == 1 and self._group_dec:
                for h in range(num_kv_heads):
                    dec_items.append((kl * 32, 1, r, h, int(qi[r])))
            else:
                for qstart in range(0, max(1, ql * group), 256):
                    for h in range(num_kv_heads):
                        pf_items.append((kl * 256, 0, r, qstart, h))
        pf_items.sort(key=lambda t: -t[0])
        dec_items.sort(key=lambda t: -t[0])
        items = pf_items + dec_items
        self._n_pf_items = len(pf_items)
        pf_cost = sum(t[0] for t in pf_items)
        dec_cost = sum(t[0] for t in dec_items)
        self._items = torch.tensor([it[1:] for it in items],
                                   dtype=torch.int32).to(dev).reshape(-1, 4)
        self._queue_head = torch.zeros(2, dtype=torch.int32, device=dev)
        self._qo_indptr_d = qo_indptr.to(dev, torch.int32)
        self._kv_indptr_d = kv_indptr.to(dev, torch.int32)
        self._kv_indices_d = kv_indices.to(dev, torch.int32)
        self._last_page_len_d = (
            (kv_len_arr.to(torch.int64) - 1) % page_size + 1
        ).to(dev, torch.int32)
        # measured routing (profiles/README r02 holistic): the persistent
        # kernel's register-allocation tax makes it ~0.7x of the composed
        # two-kernel path when decode cost dominates; the standalone MFMA
        # decode kernel is strictly faster per item. Route decode-heavy
        # batches through the composed path (still one plan/run call).
        if dec_items and dec_cost > 0.25 * (pf_cost + dec_cost):
            self._persistent = False
            if self._wrapper is None:
                ws = torch.empty(128 << 20, dtype=torch.uint8, device=dev)
                self._wrapper = BatchPrefillWithPagedKVCacheWrapper(
                    ws, self._kv_layout)
            last_page_len = ((kv_len_arr.to(torch.int64) - 1) % page_size + 1
                             ).to(torch.int32)
            self._wrapper.plan(
                qo_indptr, kv_indptr, kv_indices, last_page_len,
                num_qo_heads, num_kv_heads, head_dim_qk, page_size,
                causal=causal, sm_scale=sm_scale,
                logits_soft_cap=logits_soft_cap, q_data_type=q_data_type)
            return
        n_cu = torch.cuda.get_device_properties(dev).multi_processor_count
        self._n_wgs = min(len(items), n_cu)
        if dec_items and pf_items:
            frac = dec_cost / max(1, pf_cost + dec_cost)
            self._n_dec_wgs = max(1, min(self._n_wgs - 1,
                                         round(self._n_wgs * frac)))
        elif dec_items:
            self._n_dec_wgs = self._n_wgs
        else:
            self._n_dec_wgs = 0
        self._pi = dict(
            nnz_q=int(qi[-1]), num_qo_heads=num_qo_heads, causal=causal,
            head_dim_vo=head_dim_vo,
            sm_scale=sm_scale if sm_scale is not None
            else default_sm_scale(head_dim_qk),
            logits_soft_cap=float(logits_soft_cap or 0.0))

    def run(
        self, q: torch.Tensor, kv_cache, out: Optional[torch.Tensor] = None,
        lse: Optional[torch.Tensor] = None, return_lse: bool = True, **kwargs,
    ):
        if not self._persistent:
            return self._wrapper.run(q, kv_cache, out=out, lse=lse,
                                     return_lse=return_lse)
        from ._lib import get_ext
        from .utils import layout_code, unpack_paged_kv_cache

        pi = self._pi
        k_cache, v_cache = unpack_paged_kv_cache(kv_cache, self._kv_layout)
        if out is None:
            out = torch.empty(q.shape[0], q.shape[1], pi["head_dim_vo"],
                              dtype=q.dtype, device=q.device)
        if return_lse and lse is None:
            lse = torch.empty(q.shape[0], q.shape[1], dtype=torch.float32,
                              device=q.device)
        self._queue_head.zero_()
        get_ext().batch_attention_run(
            q, k_cache, v_cache, self._qo_indptr_d, self._kv_indices_d,
            self._kv_indptr_d, self._last_page_len_d,
            layout_code(self._kv_layout), self._items, self._queue_head,
            out, lse if return_lse else None, pi["sm_scale"],
            pi["logits_soft_cap"], -1, pi["causal"], self._group_dec,
            self._n_wgs, self._n_pf_items, self._n_dec_wgs)
        return (out, lse) if return_lse else out


class PODWithPagedKVCacheWrapper:
    r"""Prefill-On-Decode (reference pod.py:61): one prefill request runs
    CONCURRENTLY with a decode batch. The reference fuses both into one
    persistent kernel; on the 256-CU / 8-XCD chip the same effect falls out
    of two HIP streams — the MFMA-bound prefill grid and the HBM-bound
    decode grid co-schedule across CUs, so decode hides under prefill."""

    def __init__(self, float_workspace_buffer: torch.Tensor, kv_layout: str = "NHD",
                 **kwargs):
        from .decode import BatchDecodeWithPagedKVCacheWrapper

        self._decode = BatchDecodeWithPagedKVCacheWrapper(
            float_workspace_buffer, kv_layout
        )
        self._kv_layout = kv_layout
        self._aux_stream = torch.cuda.Stream(
            device=float_workspace_buffer.device
        ) if float_workspace_buffer.is_cuda else None

    def plan(self, indptr, indices, last_page_len, num_qo_heads, num_kv_heads,
             head_dim, page_size, **kwargs):
        self._decode.plan(indptr, indices, last_page_len, num_qo_heads,
                          num_kv_heads, head_dim, page_size, **kwargs)

    def run(self, q_p, k_p, v_p, q_d, paged_kv_cache, causal_p: bool = True, **kwargs):
        r"""q_p/k_p/v_p: the prefill request (contiguous KV); q_d: decode
        queries [batch, H, D] over the paged cache. The decode batch is
        issued on a second stream so both grids occupy the chip at once."""
        from .prefill import single_prefill_with_kv_cache

        main = torch.cuda.current_stream(q_p.device)
        self._aux_stream.wait_stream(main)
        with torch.cuda.stream(self._aux_stream):
            o_d = self._decode.run(q_d, paged_kv_cache)
        o_p = single_prefill_with_kv_cache(q_p, k_p, v_p, causal=causal_p)
        main.wait_stream(self._aux_stream)
        return o_p, o_d


class BatchPODWithPagedKVCacheWrapper:
    r"""Batched Prefill-On-Decode (reference pod.py
    BatchPODWithPagedKVCacheWrapper:732): a paged prefill batch and a paged
    decode batch execute concurrently — the decode grid on a second HIP
    stream co-resident with the MFMA prefill grid across the 256 CUs."""

    def __init__(self, float_workspace_buffer_p: torch.Tensor,
                 float_workspace_buffer_d: torch.Tensor,
                 kv_layout: str = "NHD", use_tensor_cores: bool = True,
                 **kwargs):
        from .decode import BatchDecodeWithPagedKVCacheWrapper
        from .prefill import BatchPrefillWithPagedKVCacheWrapper

        self._prefill = BatchPrefillWithPagedKVCacheWrapper(
            float_workspace_buffer_p, kv_layout)
        self._decode = BatchDecodeWithPagedKVCacheWrapper(
            float_workspace_buffer_d, kv_layout)
        self.device = float_workspace_buffer_p.device
        self._aux_stream = torch.cuda.Stream(device=self.device) \
            if float_workspace_buffer_p.is_cuda else None

    def plan(self, qo_indptr_p, kv_indptr_p, kv_indices_p, last_page_len_p,
             qo_indptr_d, kv_indptr_d, kv_indices_d, last_page_len_d,
             num_qo_heads, num_kv_heads, head_dim, page_size,
             pos_encoding_mode: str = "NONE", window_left: int = -1,
             q_data_type=torch.bfloat16, kv_data_type=None, data_type=None,
             sm_scale=None, rope_scale=None, rope_theta=None,
             non_blocking: bool = True, causal_p: bool = False, **kwargs):
        self._prefill.plan(
            qo_indptr_p, kv_indptr_p, kv_indices_p, last_page_len_p,
            num_qo_heads, num_kv_heads, head_dim, page_size, causal=causal_p,
            pos_encoding_mode=pos_encoding_mode, sm_scale=sm_scale,
            window_left=window_left, q_data_type=q_data_type,
            kv_data_type=kv_data_type, non_blocking=non_blocking)
        self._decode.plan(
            kv_indptr_d, kv_indices_d, last_page_len_d, num_qo_heads,
            num_kv_heads, head_dim, page_size, window_left=window_left,
            q_data_type=q_data_type, kv_data_type=kv_data_type,
            non_blocking=non_blocking)

    begin_forward = plan

    def run(self, q_p, paged_kv_cache_p, q_d, paged_kv_cache_d,
            custom_mask_p=None, packed_custom_mask_p=None,
            causal_p: bool = False, q_scale=None, k_scale=None, v_scale=None,
            return_lse: bool = False, **kwargs):
        main = torch.cuda.current_stream(q_p.device)
        self._aux_stream.wait_stream(main)
        with torch.cuda.stream(self._aux_stream):
            res_d = self._decode.run(q_d, paged_kv_cache_d,
                                     k_scale=k_scale, v_scale=v_scale,
                                     return_lse=return_lse)
        res_p = self._prefill.run(q_p, paged_kv_cache_p,
                                  return_lse=return_lse)
        main.wait_stream(self._aux_stream)
        return res_p, res_d

    forward = run

    def end_forward(self):
        pass


class BatchAttentionWithAttentionSinkWrapper(BatchAttention):
    r"""Holistic mixed-batch attention with attention sinks (reference
    flashinfer/attention BatchAttentionWithAttentionSinkWrapper role): a
    per-head virtual logit joins the softmax denominator only —
    ``out *= 1 / (1 + 2^(sink*log2(e) - lse))`` applied on the kernel's
    base-2 LSE."""

    def __init__(self, *args, sink: Optional[torch.Tensor] = None, **kwargs):
        super().__init__(*args, **kwargs)
        self._sink = sink

    def plan(self, *args, sink: Optional[torch.Tensor] = None, **kwargs):
        if sink is not None:
            self._sink = sink
        kwargs.pop("use_profiler", None)
        return super().plan(*args, **kwargs)

    def run(self, q, kv_cache, out=None, lse=None, return_lse: bool = True,
            sinks: Optional[torch.Tensor] = None, **kwargs):
        s = sinks if sinks is not None else self._sink
        o, l2 = super().run(q, kv_cache, out=out, lse=lse, return_lse=True)
        if s is not None:
            log2e = 1.4426950408889634
            corr = 1.0 / (1.0 + torch.exp2(s.float()[None, :] * log2e - l2))
            o = (o.float() * corr[..., None]).to(o.dtype)
        return (o, l2) if return_lse else o
