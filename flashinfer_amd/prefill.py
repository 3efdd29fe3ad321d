"""Prefill attention: single API + batch plan/run wrappers over paged/ragged
KV. Reference parity: flashinfer/prefill.py (single_prefill_with_kv_cache:1186,
BatchPrefillWithPagedKVCacheWrapper:1538, BatchPrefillWithRaggedKVCacheWrapper:
3193). One hand-written CDNA4 MFMA kernel (csrc/attention/batch_prefill.hip)
serves all three — GQA rows packed (q_pos x group), CTA tile 128 rows.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple, Union

import torch

from ._lib import get_ext
from .api_logging import flashinfer_api
from .fi_trace import fi_trace
from .utils import ceil_div, default_sm_scale, layout_code, unpack_paged_kv_cache

def _plan_tiles(qo_lens, group: int, causal: bool = False):
    """Pick the q-tile size (128 or 256 packed rows) from the average packed
    length, then emit (req, qstart) work items (reference scheduler.cuh:616
    role — CTA_TILE_Q from avg packed qo_len). For causal attention tiles are
    emitted LONGEST-FIRST (cost grows with qstart) so the hardware dispatch
    order doesn't leave the expensive diagonal tiles for the tail."""
    packed = [L * group for L in qo_lens]
    avg = sum(packed) / max(1, len(packed))
    cta_q = 256 if avg >= 192 else 128
    items = []
    for b, pk in enumerate(packed):
        for start in range(0, max(pk, 1), cta_q):
            items.append((b, start))
    if causal:
        items.sort(key=lambda t: -t[1])
    tile_req = [b for b, _ in items]
    tile_qstart = [st for _, st in items]
    return cta_q, tile_req, tile_qstart


# split-KV engages when the (q_tile x kv_head) grid underfills the 256-CU
# chip (reference scheduler.cuh:545 PrefillSplitQOKVIndptr + :101 binary
# search role): a bs=1 long-kv short-q prefill otherwise runs on a handful
# of workgroups.
_SPLIT_TARGET_UNITS = 512
_SPLIT_MIN_CHUNK = 256  # tokens (multiple of the 64-token kv tile)


def _plan_tiles_split(qo_lens, kv_lens, group, causal, num_kv_heads,
                      num_qo_heads, head_dim_vo, ws_bytes):
    """Emit (req, qstart[, kv_chunk]) work items, splitting KV when the base
    grid underfills the chip. Returns a dict with the tile arrays and (when
    split) the merge bookkeeping."""
    cta_q, tile_req, tile_qstart = _plan_tiles(qo_lens, group, causal)
    units = len(tile_req) * num_kv_heads
    max_kv = max(kv_lens) if kv_lens else 0
    if units >= _SPLIT_TARGET_UNITS or max_kv < 2 * _SPLIT_MIN_CHUNK:
        return dict(cta_q=cta_q, tile_req=tile_req, tile_qstart=tile_qstart,
                    split=False)
    want = min(-(-_SPLIT_TARGET_UNITS // max(1, units)),
               -(-max_kv // _SPLIT_MIN_CHUNK))
    kv_chunk = -(-max_kv // want)
    kv_chunk = -(-kv_chunk // _SPLIT_MIN_CHUNK) * _SPLIT_MIN_CHUNK
    # bound the partial buffers by the float workspace
    def slots_for(ck):
        return sum(q * max(1, -(-k // ck)) for q, k in zip(qo_lens, kv_lens))
    bytes_per_slot = num_qo_heads * (head_dim_vo + 1) * 4
    while slots_for(kv_chunk) * bytes_per_slot > ws_bytes * 0.9:
        kv_chunk *= 2
        if kv_chunk >= max_kv:
            return dict(cta_q=cta_q, tile_req=tile_req,
                        tile_qstart=tile_qstart, split=False)
    n_chunks = [max(1, -(-k // kv_chunk)) for k in kv_lens]
    slot_base, acc = [], 0
    for q, nc in zip(qo_lens, n_chunks):
        slot_base.append(acc)
        acc += q * nc
    n_slots = acc
    # every (qtile, chunk) pair is emitted — chunks fully beyond a causal
    # bound early-out in-kernel and write empty partials, which the LSE
    # merge treats as zero weight (no uninitialized slots by construction)
    s_req, s_qstart, s_chunk = [], [], []
    for b, st in zip(tile_req, tile_qstart):
        for c in range(n_chunks[b]):
            s_req.append(b)
            s_qstart.append(st)
            s_chunk.append(c)
    merge_indptr = [0]
    for q, nc in zip(qo_lens, n_chunks):
        for _ in range(q):
            merge_indptr.append(merge_indptr[-1] + nc)
    return dict(cta_q=cta_q, tile_req=s_req, tile_qstart=s_qstart,
                tile_kv_chunk=s_chunk, kv_chunk=kv_chunk,
                slot_base=slot_base, n_slots=n_slots,
                merge_indptr=merge_indptr, split=True)


class _BatchPrefillBase:
    def __init__(self, float_workspace_buffer, kv_layout="NHD", use_cuda_graph=False,
                 backend="fa2", jit_args=None, jit_module=None, **kwargs):
        self._float_workspace_buffer = float_workspace_buffer
        self.device = float_workspace_buffer.device
        self._kv_layout = kv_layout
        self._use_cuda_graph = use_cuda_graph
        self._plan_info = None
        # JIT attention variant (reference customize-config mechanism):
        # a module from jit.attention.gen_customize_batch_prefill_module
        self._custom_fn = jit_module.run_ptr if jit_module is not None else 0

    @property
    def is_cuda_graph_enabled(self):
        return self._use_cuda_graph

    def reset_workspace_buffer(self, float_workspace_buffer, int_workspace_buffer=None):
        self._float_workspace_buffer = float_workspace_buffer

    def _pack_mask(self, custom_mask, packed_custom_mask, qo_lens, kv_lens):
        """Pack (or accept pre-packed) per-request [qo_len, kv_len] boolean
        masks into byte-aligned little-endian segments (reference
        quantization.packbits mechanism)."""
        self._mask_data = None
        self._mask_byte_indptr = None
        if custom_mask is None and packed_custom_mask is None:
            return
        from .quantization import segment_packbits

        qk_lens = [q * k for q, k in zip(qo_lens, kv_lens)]
        if packed_custom_mask is not None:
            byte_indptr = torch.zeros(len(qk_lens) + 1, dtype=torch.int32)
            byte_indptr[1:] = torch.cumsum(
                torch.tensor([ceil_div(L, 8) for L in qk_lens]), 0
            ).int()
            self._mask_data = packed_custom_mask.to(self.device)
            self._mask_byte_indptr = byte_indptr.to(self.device)
            return
        qk_indptr = torch.zeros(len(qk_lens) + 1, dtype=torch.int64)
        qk_indptr[1:] = torch.cumsum(torch.tensor(qk_lens), 0)
        packed, byte_indptr = segment_packbits(
            custom_mask.reshape(-1).to(self.device), qk_indptr.to(self.device)
        )
        self._mask_data = packed
        self._mask_byte_indptr = byte_indptr.to(torch.int32)

    def _plan_common(self, qo_indptr, num_qo_heads, num_kv_heads, head_dim, causal,
                     sm_scale, window_left, logits_soft_cap, non_blocking=True,
                     alibi=False, kv_lens=None, head_dim_vo=None):
        qi = qo_indptr.to("cpu", torch.int64)
        qo_lens = (qi[1:] - qi[:-1]).tolist()
        group = num_qo_heads // num_kv_heads
        head_dim_vo = head_dim_vo or head_dim
        plan = _plan_tiles_split(
            qo_lens, kv_lens if kv_lens is not None else [0] * len(qo_lens),
            group, causal, num_kv_heads, num_qo_heads, head_dim_vo,
            self._float_workspace_buffer.numel()
            * self._float_workspace_buffer.element_size(),
        )
        cta_q = plan["cta_q"]
        tile_req, tile_qstart = plan["tile_req"], plan["tile_qstart"]
        n_tiles = len(tile_req)
        self._split = plan["split"]
        if self._split:
            meta = torch.tensor(
                tile_req + tile_qstart + plan["tile_kv_chunk"] + plan["slot_base"]
                + plan["merge_indptr"], dtype=torch.int32,
            ).to(self.device, non_blocking=non_blocking)
            self._tile_req = meta[:n_tiles]
            self._tile_qstart = meta[n_tiles : 2 * n_tiles]
            self._tile_kv_chunk = meta[2 * n_tiles : 3 * n_tiles]
            nb = len(plan["slot_base"])
            self._req_slot_base = meta[3 * n_tiles : 3 * n_tiles + nb]
            self._merge_indptr = meta[3 * n_tiles + nb :]
            self._kv_chunk = plan["kv_chunk"]
            from .utils import WorkspaceAllocator

            alloc = WorkspaceAllocator(self._float_workspace_buffer)
            ns = plan["n_slots"]
            self._tmp_v = alloc.alloc(
                ns * num_qo_heads * head_dim_vo * 4, torch.float32,
                (ns, num_qo_heads, head_dim_vo))
            self._tmp_s = alloc.alloc(
                ns * num_qo_heads * 4, torch.float32, (ns, num_qo_heads))
        else:
            # causal short-diagonal retiling: tiles whose causal KV reach is
            # < 512 tokens are dominated by the fixed CTAQ-256 Q-stage /
            # epilogue cost (measured: causal 374 TF vs 514 non-causal at
            # bs=16 s=1024 with only ~6% diagonal MFMA waste — profiles/
            # README r02 addendum 7). They are re-emitted as CTAQ-128
            # tiles and run as a second launch, halving their fixed cost
            # and diagonal waste.
            self._tile_req2 = None
            self._tile_chain_req = None
            if causal and cta_q == 256 and kv_lens is not None:
                big_r, big_s, big_kv = [], [], []
                sm_r, sm_s = [], []
                shorts = []   # ORIGINAL 256-row short tiles (for chaining)
                for b, st in zip(tile_req, tile_qstart):
                    pk = qo_lens[b] * group
                    kvh = kv_lens[b] - qo_lens[b] + (min(st + cta_q, pk)
                                                     + group - 1) // group
                    if kvh < 512:
                        shorts.append((b, st, kvh))
                        sm_r.append(b)          # 128-row retiling for the
                        sm_s.append(st)         # minority second-launch path
                        if st + 128 < pk:
                            sm_r.append(b)
                            sm_s.append(st + 128)
                    else:
                        big_r.append(b)
                        big_s.append(st)
                        big_kv.append(kvh)
                # only worth it when short tiles are a MINORITY: when most
                # tiles are short (s ~ 1024 at group 4, or group-1 ragged)
                # doubling their count costs more in tail rounds than the
                # halved fixed cost saves (measured: bs=1/s=8192 439->493
                # TF, bs=4/s=4096 551 TF; bs=16/s=1024 374->368 and
                # ragged-192 442->418 without this gate)
                if sm_r and big_r and len(sm_r) <= len(big_r) // 2:
                    tile_req, tile_qstart = big_r, big_s
                    n_tiles = len(tile_req)
                    meta2 = torch.tensor(sm_r + sm_s, dtype=torch.int32).to(
                        self.device, non_blocking=non_blocking)
                    self._tile_req2 = meta2[: len(sm_r)]
                    self._tile_qstart2 = meta2[len(sm_r):]
                elif (len(shorts) >= 4 and not getattr(self, "_custom_fn", 0)):
                    # shorts-majority (e.g. bs=16 s=1024): CHAIN pairs of
                    # short 256-row tiles into one WG instead — halves the
                    # WG count for the short population (fewer dispatch
                    # rounds) with no extra fixed cost. Pair smallest-with-
                    # largest so chained costs balance.
                    order = sorted(range(len(shorts)), key=lambda i: shorts[i][2])
                    pr_r, pr_s, ch_r, ch_s, costs = [], [], [], [], []
                    lo, hi = 0, len(order) - 1
                    while lo < hi:
                        a, z = order[hi], order[lo]   # big + small
                        pr_r.append(shorts[a][0]); pr_s.append(shorts[a][1])
                        ch_r.append(shorts[z][0]); ch_s.append(shorts[z][1])
                        costs.append(shorts[a][2] + shorts[z][2])
                        lo += 1; hi -= 1
                    if lo == hi:
                        a = order[lo]
                        pr_r.append(shorts[a][0]); pr_s.append(shorts[a][1])
                        ch_r.append(-1); ch_s.append(-1)
                        costs.append(shorts[a][2])
                    items = sorted(
                        list(zip(big_r, big_s, [-1] * len(big_r),
                                 [-1] * len(big_r), big_kv))
                        + list(zip(pr_r, pr_s, ch_r, ch_s, costs)),
                        key=lambda t: -t[4])
                    tile_req = [t[0] for t in items]
                    tile_qstart = [t[1] for t in items]
                    n_tiles = len(tile_req)
                    metac = torch.tensor(
                        [t[2] for t in items] + [t[3] for t in items],
                        dtype=torch.int32).to(self.device,
                                              non_blocking=non_blocking)
                    self._tile_chain_req = metac[:n_tiles]
                    self._tile_chain_qstart = metac[n_tiles:]
            meta = torch.tensor(tile_req + tile_qstart, dtype=torch.int32).to(
                self.device, non_blocking=non_blocking
            )
            self._tile_req = meta[:n_tiles]
            self._tile_qstart = meta[n_tiles:]
        self._qo_indptr_d = qo_indptr.to(self.device, torch.int32,
                                         non_blocking=non_blocking)
        self._plan_info = dict(
            num_qo_heads=num_qo_heads, num_kv_heads=num_kv_heads, head_dim=head_dim,
            head_dim_vo=head_dim_vo,
            causal=causal, window_left=window_left,
            logits_soft_cap=float(logits_soft_cap or 0.0),
            sm_scale=sm_scale if sm_scale is not None else default_sm_scale(head_dim),
            nnz_q=int(qi[-1]), cta_q=cta_q, alibi=alibi,
        )

    def _run_common(self, q, k_cache, v_cache, kv_indices, kv_indptr, kv_last_page_len,
                    paged, out, lse, return_lse, k_scale=None, v_scale=None,
                    profiler_buffer=None):
        pi = self._plan_info
        if pi is None:
            raise RuntimeError("must call plan() before run()")
        vo = pi.get("head_dim_vo") or q.shape[2]
        if out is None:
            out = torch.empty(q.shape[0], q.shape[1], vo, dtype=q.dtype,
                              device=q.device)
        split = getattr(self, "_split", False)
        if (return_lse or split) and lse is None:
            lse = torch.empty(q.shape[0], q.shape[1], dtype=torch.float32,
                              device=q.device)
        sm_scale = pi["sm_scale"]
        kv_fp8 = k_cache.dtype in (torch.float8_e4m3fn, torch.uint8)
        k_descale = v_descale = 1.0
        if kv_fp8:
            # fp8 KV: scales are applied in the dequantizing LDS staging
            k_descale = k_scale if k_scale is not None else 1.0
            v_descale = v_scale if v_scale is not None else 1.0
        elif k_scale is not None:
            sm_scale *= k_scale
        get_ext().batch_prefill_run(
            q, k_cache, v_cache, self._qo_indptr_d, kv_indices, kv_indptr,
            kv_last_page_len, layout_code(self._kv_layout), self._tile_req,
            self._tile_qstart, out, lse if return_lse else None, sm_scale,
            pi["logits_soft_cap"], pi["window_left"], pi["causal"], paged,
            pi["cta_q"], getattr(self, "_mask_data", None),
            getattr(self, "_mask_byte_indptr", None), pi.get("alibi", False),
            k_descale, v_descale, profiler_buffer,
            self._tile_kv_chunk if split else None,
            self._kv_chunk if split else 0,
            self._req_slot_base if split else None,
            self._tmp_v if split else None,
            self._tmp_s if split else None,
            getattr(self, "_custom_fn", 0),
            getattr(self, "_tile_chain_req", None) if not split else None,
            getattr(self, "_tile_chain_qstart", None) if not split else None,
        )
        if getattr(self, "_tile_req2", None) is not None and not split:
            # second launch: the causal short-diagonal tiles at CTAQ 128
            get_ext().batch_prefill_run(
                q, k_cache, v_cache, self._qo_indptr_d, kv_indices, kv_indptr,
                kv_last_page_len, layout_code(self._kv_layout),
                self._tile_req2, self._tile_qstart2, out,
                lse if return_lse else None, sm_scale,
                pi["logits_soft_cap"], pi["window_left"], pi["causal"], paged,
                128, getattr(self, "_mask_data", None),
                getattr(self, "_mask_byte_indptr", None),
                pi.get("alibi", False), k_descale, v_descale, profiler_buffer,
                None, 0, None, None, None, getattr(self, "_custom_fn", 0),
                None, None,
            )
        if split:
            # LSE merge of the per-chunk partials (cascade merge kernel)
            get_ext().merge_states(
                self._tmp_v, self._tmp_s, out, lse, self._merge_indptr, 0,
                q.shape[0])
        if not kv_fp8 and v_scale is not None:
            out = out * v_scale
        return (out, lse) if return_lse else out

    def end_forward(self):
        pass


class BatchPrefillWithPagedKVCacheWrapper(_BatchPrefillBase):
    r"""Batch prefill/append attention over a paged KV cache (plan/run)."""

    def plan(
        self, qo_indptr, paged_kv_indptr, paged_kv_indices, paged_kv_last_page_len,
        num_qo_heads, num_kv_heads, head_dim_qk, page_size,
        head_dim_vo=None, custom_mask=None, packed_custom_mask=None,
        causal: bool = False, pos_encoding_mode: str = "NONE",
        sm_scale=None, window_left: int = -1, logits_soft_cap=None,
        rope_scale=None, rope_theta=None, q_data_type=torch.bfloat16,
        kv_data_type=None, o_data_type=None, non_blocking: bool = True, **kwargs,
    ):
        if pos_encoding_mode not in ("NONE", "ALIBI"):
            raise NotImplementedError("apply RoPE beforehand")
        ip = paged_kv_indptr.to("cpu", torch.int64)
        lp = paged_kv_last_page_len.to("cpu", torch.int64)
        np_ = ip[1:] - ip[:-1]
        kv_lens = (torch.clamp(np_ - 1, min=0) * page_size
                   + torch.where(np_ > 0, lp, torch.zeros_like(lp))).tolist()
        self._plan_common(qo_indptr, num_qo_heads, num_kv_heads, head_dim_qk, causal,
                          sm_scale, window_left, logits_soft_cap, non_blocking,
                          alibi=pos_encoding_mode == "ALIBI", kv_lens=kv_lens,
                          head_dim_vo=head_dim_vo)
        self._kv_indptr_d = paged_kv_indptr.to(self.device, torch.int32,
                                               non_blocking=non_blocking)
        self._kv_indices_d = paged_kv_indices.to(self.device, torch.int32,
                                                 non_blocking=non_blocking)
        self._kv_last_page_len_d = paged_kv_last_page_len.to(
            self.device, torch.int32, non_blocking=non_blocking)
        if custom_mask is not None or packed_custom_mask is not None:
            qi = qo_indptr.to("cpu", torch.int64)
            qo_lens = (qi[1:] - qi[:-1]).tolist()
            self._pack_mask(custom_mask, packed_custom_mask, qo_lens, kv_lens)

    begin_forward = plan

    def run(self, q, paged_kv_cache, *args, k_scale=None, v_scale=None, out=None,
            lse=None, return_lse: bool = False, profiler_buffer=None, **kwargs):
        k_cache, v_cache = unpack_paged_kv_cache(paged_kv_cache, self._kv_layout)
        return self._run_common(q, k_cache, v_cache, self._kv_indices_d,
                                self._kv_indptr_d, self._kv_last_page_len_d, True,
                                out, lse, return_lse, k_scale, v_scale,
                                profiler_buffer)

    forward = run


class BatchPrefillWithRaggedKVCacheWrapper(_BatchPrefillBase):
    r"""Batch prefill attention over contiguous (ragged) KV (plan/run)."""

    def plan(
        self, qo_indptr, kv_indptr, num_qo_heads, num_kv_heads, head_dim_qk,
        head_dim_vo=None, custom_mask=None, packed_custom_mask=None,
        causal: bool = False, pos_encoding_mode: str = "NONE",
        sm_scale=None, window_left: int = -1, logits_soft_cap=None,
        rope_scale=None, rope_theta=None, q_data_type=torch.bfloat16,
        kv_data_type=None, o_data_type=None, non_blocking: bool = True, **kwargs,
    ):
        if pos_encoding_mode not in ("NONE", "ALIBI"):
            raise NotImplementedError("apply RoPE beforehand")
        ki = kv_indptr.to("cpu", torch.int64)
        kv_lens = (ki[1:] - ki[:-1]).tolist()
        self._plan_common(qo_indptr, num_qo_heads, num_kv_heads, head_dim_qk, causal,
                          sm_scale, window_left, logits_soft_cap, non_blocking,
                          alibi=pos_encoding_mode == "ALIBI", kv_lens=kv_lens,
                          head_dim_vo=head_dim_vo)
        self._kv_indptr_d = kv_indptr.to(self.device, torch.int32,
                                         non_blocking=non_blocking)
        if custom_mask is not None or packed_custom_mask is not None:
            qi = qo_indptr.to("cpu", torch.int64)
            ki = kv_indptr.to("cpu", torch.int64)
            self._pack_mask(custom_mask, packed_custom_mask,
                            (qi[1:] - qi[:-1]).tolist(), (ki[1:] - ki[:-1]).tolist())

    begin_forward = plan

    def run(self, q, k, v, *args, k_scale=None, v_scale=None, out=None, lse=None,
            return_lse: bool = False, profiler_buffer=None, **kwargs):
        return self._run_common(q, k, v, None, self._kv_indptr_d, None, False,
                                out, lse, return_lse, k_scale, v_scale,
                                profiler_buffer)

    forward = run


@flashinfer_api
@fi_trace
def single_prefill_with_kv_cache(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    custom_mask=None, packed_custom_mask=None, causal: bool = False,
    kv_layout: str = "NHD", pos_encoding_mode: str = "NONE",
    use_fp16_qk_reduction: bool = False, sm_scale=None, window_left: int = -1,
    logits_soft_cap=None, rope_scale=None, rope_theta=None,
    return_lse: bool = False, backend: str = "auto", **kwargs,
):
    r"""Prefill attention for one request.

    q: [qo_len, Hq, D]; k/v: [kv_len, Hkv, D] (NHD) or [Hkv, kv_len, D] (HND).
    """
    if pos_encoding_mode not in ("NONE", "ALIBI"):
        raise NotImplementedError("apply RoPE beforehand")
    if kv_layout == "HND":
        k = k.transpose(0, 1)
        v = v.transpose(0, 1)
    qo_len, Hq, D = q.shape
    kv_len, Hkv, _ = k.shape
    Dvo = v.shape[-1]
    dev = q.device
    group = Hq // Hkv
    cta_q, tile_req, tile_qstart = _plan_tiles([qo_len], group, causal)
    n_tiles = len(tile_req)
    meta = torch.tensor(
        tile_req + tile_qstart + [0, qo_len, 0, kv_len], dtype=torch.int32
    ).to(dev)
    out = torch.empty(qo_len, Hq, Dvo, dtype=q.dtype, device=dev)
    lse = torch.empty(qo_len, Hq, dtype=torch.float32, device=dev) if return_lse else None
    mask_data = mask_indptr = None
    if custom_mask is not None or packed_custom_mask is not None:
        if packed_custom_mask is None:
            from .quantization import packbits

            packed_custom_mask = packbits(custom_mask.reshape(-1))
        mask_data = packed_custom_mask.to(dev)
        mask_indptr = torch.zeros(2, dtype=torch.int32, device=dev)
    get_ext().batch_prefill_run(
        q, k, v, meta[2 * n_tiles : 2 * n_tiles + 2], None,
        meta[2 * n_tiles + 2 :], None, 0, meta[:n_tiles],
        meta[n_tiles : 2 * n_tiles], out, lse,
        sm_scale if sm_scale is not None else default_sm_scale(D),
        float(logits_soft_cap or 0.0), window_left, causal, False, cta_q,
        mask_data, mask_indptr, pos_encoding_mode == "ALIBI", 1.0, 1.0, None,
        None, 0, None, None, None, 0, None, None,
    )
    return (out, lse) if return_lse else out


def single_prefill_with_kv_cache_return_lse(*args, **kwargs):
    r"""Reference-name alias: ``single_prefill_with_kv_cache(...,
    return_lse=True)``."""
    kwargs["return_lse"] = True
    return single_prefill_with_kv_cache(*args, **kwargs)
