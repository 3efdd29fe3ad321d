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


@pytest.mark.gpu
def test_deepseek_mla_moe_example():
    import importlib.util
    from pathlib import Path

    path = Path(__file__).resolve().parent.parent / "examples" / \
        "deepseek_mla_moe_serving.py"
    spec = importlib.util.spec_from_file_location("dsv3_example", path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod.main(batch=4, ctx_len=128, steps=2, layers=1)
