"""XQA speculative-decode attention (reference parity: flashinfer/xqa.py
xqa:188, xqa_mla:~400 — the trtllm XQA kernel's serving contract). On MI355X
the same contract lowers onto the two CDNA4 attention kernels: q_seq_len == 1
runs the wide-vector decode kernel, q_seq_len > 1 (speculative tokens,
right-aligned causal) runs the MFMA prefill kernel; fp8 KV caches use the
kernels' native dequant paths."""
from __future__ import annotations

import math
from typing import Optional, Union

import torch


def _plan_inputs(page_table, seq_lens, page_size):
    B = seq_lens.numel()
    lens = seq_lens.to("cpu", torch.int64)
    pages_per = (lens + page_size - 1) // page_size
    kv_indptr = torch.zeros(B + 1, dtype=torch.int32)
    kv_indptr[1:] = pages_per.cumsum(0).int()
    pt = page_table.to("cpu", torch.int64)
    kv_indices = torch.cat(
        [pt[b, : int(pages_per[b])] for b in range(B)]).to(torch.int32)
    last = ((lens - 1) % page_size + 1).to(torch.int32)
    last = torch.where(lens > 0, last, torch.zeros_like(last))
    return kv_indptr, kv_indices, last


def xqa(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    page_table: torch.Tensor,
    seq_lens: torch.Tensor,
    output: torch.Tensor,
    workspace_buffer: torch.Tensor,
    semaphores: Optional[torch.Tensor] = None,
    num_kv_heads: int = 0,
    page_size: int = 0,
    sinks: Optional[torch.Tensor] = None,
    q_scale: Union[float, torch.Tensor] = 1.0,
    kv_scale: Union[float, torch.Tensor] = 1.0,
    sliding_win_size: int = 0,
    kv_layout: str = "NHD",
    sm_count: Optional[int] = None,
    enable_pdl: Optional[bool] = None,
    rcp_out_scale: float = 1.0,
    q_seq_len: int = 1,
    mask: Optional[torch.Tensor] = None,
    *,
    q_cu_seq_lens: Optional[torch.Tensor] = None,
    k_sf_cache=None, v_sf_cache=None,
) -> None:
    r"""q: [B, 1(beam), Hq, D] (decode) or [B, 1, q_seq_len, Hq, D]
    (speculative); writes ``output`` (same shape) in place. fp8 e4m3 KV uses
    ``kv_scale`` as the dequant factor."""
    from .decode import BatchDecodeWithPagedKVCacheWrapper
    from .prefill import BatchPrefillWithPagedKVCacheWrapper

    spec = q.dim() == 5
    B = q.shape[0]
    if q.shape[1] != 1:
        raise NotImplementedError("beam_width > 1 not supported")
    Hq, D = q.shape[-2], q.shape[-1]
    if page_size == 0:
        page_size = k_cache.shape[1] if kv_layout == "NHD" else k_cache.shape[2]
    if num_kv_heads == 0:
        num_kv_heads = k_cache.shape[2] if kv_layout == "NHD" else k_cache.shape[1]
    qs = float(q_scale) if not torch.is_tensor(q_scale) else float(q_scale.item())
    kvs = float(kv_scale) if not torch.is_tensor(kv_scale) else float(kv_scale.item())
    sm_scale = qs / math.sqrt(D)
    window_left = sliding_win_size - 1 if sliding_win_size > 0 else -1
    kv_indptr, kv_indices, last = _plan_inputs(page_table, seq_lens, page_size)
    kv_fp8 = k_cache.dtype == torch.float8_e4m3fn

    if not spec and q_seq_len == 1 and mask is None:
        w = BatchDecodeWithPagedKVCacheWrapper(workspace_buffer, kv_layout)
        w.plan(kv_indptr, kv_indices, last, Hq, num_kv_heads, D, page_size,
               window_left=window_left, q_data_type=q.dtype,
               kv_data_type=k_cache.dtype, sm_scale=sm_scale)
        o = w.run(q.reshape(B, Hq, D), (k_cache, v_cache), sinks=sinks,
                  k_scale=kvs if kv_fp8 else None,
                  v_scale=kvs if kv_fp8 else None)
    else:
        nq = q_seq_len if spec else 1
        qo_indptr = torch.arange(0, (B + 1) * nq, nq, dtype=torch.int32)
        w = BatchPrefillWithPagedKVCacheWrapper(workspace_buffer, kv_layout)
        w.plan(qo_indptr, kv_indptr, kv_indices, last, Hq, num_kv_heads, D,
               page_size, causal=True, window_left=window_left,
               sm_scale=sm_scale, q_data_type=q.dtype,
               kv_data_type=k_cache.dtype,
               custom_mask=mask)
        o = w.run(q.reshape(B * nq, Hq, D), (k_cache, v_cache),
                  k_scale=kvs if kv_fp8 else None,
                  v_scale=kvs if kv_fp8 else None)
    if rcp_out_scale != 1.0:
        o = o * rcp_out_scale
    output.copy_(o.reshape(output.shape).to(output.dtype))


def xqa_mla(
    q: torch.Tensor,                      # [B, 1, Hq, 576] (512 ckv + 64 kpe)
    k_cache: torch.Tensor,                # [total_cache_slots, 576] (fp8 or bf16)
    v_cache: torch.Tensor,                # 512-wide view of the same cache
    page_table: torch.Tensor,
    seq_lens: torch.Tensor,
    output: torch.Tensor,                 # [B, 1, Hq, 512]
    workspace_buffer: torch.Tensor,
    semaphores: Optional[torch.Tensor] = None,
    page_size: int = 0,
    q_scale: Union[float, torch.Tensor] = 1.0,
    kv_scale: Union[float, torch.Tensor] = 1.0,
    sm_count: Optional[int] = None,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""MLA decode in XQA clothing: 576-d packed q / KV cache (512 compressed
    + 64 rope), output 512-d — runs the d-sliced MLA MFMA kernel; an fp8 KV
    cache dequantizes in the kernel's LDS staging with ``kv_scale``."""
    from .mla import BatchMLAPagedAttentionWrapper

    B, _, Hq, Dq = q.shape
    if Dq != 576:
        raise ValueError("xqa_mla expects 512+64 packed head dim")
    if page_size == 0:
        raise ValueError("page_size is required")
    kv_indptr, kv_indices, _ = _plan_inputs(page_table, seq_lens, page_size)
    qs = float(q_scale) if not torch.is_tensor(q_scale) else float(q_scale.item())
    pages = k_cache.reshape(-1, page_size, 576)
    ckv = pages[:, :, :512].contiguous()
    kpe = pages[:, :, 512:].contiguous()
    qc = q[:, 0]
    if qc.dtype == torch.float8_e4m3fn:   # our MFMA pipeline computes in bf16
        qc = qc.to(torch.bfloat16)
    w = BatchMLAPagedAttentionWrapper(workspace_buffer)
    qo_indptr = torch.arange(0, B + 1, dtype=torch.int32)
    w.plan(qo_indptr, kv_indptr, kv_indices,
           seq_lens.to(torch.int32), Hq, 512, 64, page_size, causal=False,
           sm_scale=qs * (576.0 ** -0.5), q_data_type=qc.dtype)
    kvs = float(kv_scale) if not torch.is_tensor(kv_scale) else float(kv_scale.item())
    o = w.run(qc[..., :512].contiguous(), qc[..., 512:].contiguous(),
              ckv, kpe, ckv_scale=kvs, kpe_scale=kvs)
    output.copy_(o.reshape(output.shape).to(output.dtype))
