"""GPU numerics: MFMA GEMM vs torch.matmul (random asymmetric operands —
transpose-detecting by construction, guide ERRATA #3)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("M,N,K", [
    (128, 128, 64), (256, 512, 1024), (100, 300, 72), (1, 128, 256),
    (4096, 4096, 4096),
])
def test_mm_bf16(M, N, K):
    from flashinfer_amd.gemm import mm_bf16

    torch.manual_seed(0)
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") / 8
    b = (torch.randn(N, K, dtype=torch.bfloat16, device="cuda") / 8).t()
    out = mm_bf16(a, b)
    ref = a.float() @ b.float()
    torch.testing.assert_close(out.float(), ref, atol=0.02 * K ** 0.5 / 8, rtol=2e-2)
