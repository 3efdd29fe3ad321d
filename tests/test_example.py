"""End-to-end integration: the Llama serving example generates tokens using
only flashinfer_amd kernels (attention, rope, norm, GEMM, paging, sampling)."""
import sys
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_llama_serving_example():
    sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "examples"))
    import llama_serving

    out = llama_serving.main(batch=2, prompt_len=32, gen_tokens=4, layers=2,
                             vocab=1000)
    assert out.shape == (2, 4)
