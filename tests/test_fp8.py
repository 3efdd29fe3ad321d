"""GPU tests: fp8 quantization + groupwise/per-tensor fp8 GEMM."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_per_group_quant_roundtrip():
    from flashinfer_amd.fp8_quantization import per_token_group_quant_fp8

    torch.manual_seed(0)
    x = torch.randn(64, 512, dtype=torch.bfloat16, device="cuda") * 3
    q, s = per_token_group_quant_fp8(x, transpose_scale=False)
    assert q.shape == (64, 512) and s.shape == (64, 4)
    deq = q.float() * s.repeat_interleave(128, dim=1)
    torch.testing.assert_close(deq, x.float(), atol=0.15, rtol=0.1)
    qt, st = per_token_group_quant_fp8(x, transpose_scale=True)
    assert st.shape == (4, 64)
    torch.testing.assert_close(st.t().contiguous(), s)


def test_gemm_fp8_groupwise_matches_dequant():
    from flashinfer_amd.fp8_quantization import (
        gemm_fp8_nt_groupwise,
        per_block_quant_fp8,
        per_token_group_quant_fp8,
    )

    torch.manual_seed(0)
    M, N, K = 256, 512, 384
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    a_q, a_s = per_token_group_quant_fp8(a, transpose_scale=True)
    b_q, b_s = per_block_quant_fp8(b)
    out = gemm_fp8_nt_groupwise(a_q, b_q, a_s, b_s)
    # dequantized reference
    a_deq = a_q.float() * a_s.t().repeat_interleave(128, dim=1)
    bs_full = b_s.t().repeat_interleave(128, 0)[:N].repeat_interleave(128, 1)[:, :K]
    b_deq = b_q.float() * bs_full
    ref = a_deq @ b_deq.t()
    torch.testing.assert_close(out.float(), ref, atol=0.5, rtol=5e-2)


def test_bmm_fp8():
    from flashinfer_amd.fp8_quantization import bmm_fp8

    torch.manual_seed(0)
    B, M, K, N = 4, 128, 256, 128
    a = (torch.randn(B, M, K, device="cuda") / 4).to(torch.float8_e4m3fn)
    b = (torch.randn(B, N, K, device="cuda") / 4).to(torch.float8_e4m3fn)
    b_cm = b.transpose(1, 2)  # [B, K, N] column-major
    out = bmm_fp8(a, b_cm, torch.tensor(2.0), torch.tensor(3.0))
    ref = torch.einsum("bmk,bnk->bmn", a.float(), b.float()) * 6.0
    torch.testing.assert_close(out.float(), ref, atol=0.2, rtol=5e-2)


def test_group_gemm_fp8():
    from flashinfer_amd.fp8_quantization import (
        group_gemm_fp8_nt_groupwise,
        per_block_quant_fp8,
        per_token_group_quant_fp8,
    )

    torch.manual_seed(0)
    S, N, K = 3, 256, 256
    m_sizes = [100, 0, 200]
    M = sum(m_sizes)
    m_indptr = torch.zeros(S + 1, dtype=torch.int32, device="cuda")
    m_indptr[1:] = torch.cumsum(torch.tensor(m_sizes, device="cuda"), 0).int()
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(S, N, K, dtype=torch.bfloat16, device="cuda")
    a_q, a_s = per_token_group_quant_fp8(a, transpose_scale=True)
    w_q, w_s = per_block_quant_fp8(w)
    out = group_gemm_fp8_nt_groupwise(a_q, w_q, m_indptr, a_s, w_s)
    a_deq = a_q.float() * a_s.t().repeat_interleave(128, dim=1)
    for e in range(S):
        s, t = int(m_indptr[e]), int(m_indptr[e + 1])
        if s == t:
            continue
        ws_full = w_s[e].t().repeat_interleave(128, 0)[:N].repeat_interleave(128, 1)[:, :K]
        ref = a_deq[s:t] @ (w_q[e].float() * ws_full).t()
        torch.testing.assert_close(out[s:t].float(), ref, atol=0.5, rtol=5e-2)
