"""Reference-correctness suite over the trace-template registry (reference
tests/trace/test_*_reference_correctness.py design: every template's inputs
run through the library op AND its pure-torch reference formula)."""
import pytest
import torch

from flashinfer_amd.trace import templates, validate_trace_record

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("name", sorted(templates()))
def test_template_reference_correctness(name):
    import flashinfer_amd as fi

    t = templates()[name]
    torch.manual_seed(0)
    args, kwargs = t.make_inputs(torch, "cuda")
    actual = t.run(fi, *args, **kwargs)
    expected = t.reference(torch, *args, **kwargs)
    if not isinstance(actual, tuple):
        actual, expected = (actual,), (expected,)
    for a, e in zip(actual, expected):
        if t.atol == 0:
            assert torch.equal(a, e)
        else:
            torch.testing.assert_close(a, e.to(a.dtype), atol=t.atol,
                                       rtol=t.rtol)


def test_trace_dump_validates(tmp_path, monkeypatch):
    """fi_trace dumps parse and validate against the registry."""
    import importlib
    import json
    import os

    monkeypatch.setenv("FLASHINFER_TRACE_DUMP", str(tmp_path))
    import flashinfer_amd.fi_trace as ft
    importlib.reload(ft)
    import flashinfer_amd as fi

    @ft.fi_trace
    def single_decode_with_kv_cache(q, k, v):
        return fi.single_decode_with_kv_cache(q, k, v)

    q = torch.randn(8, 128, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(64, 2, 128, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(64, 2, 128, dtype=torch.bfloat16, device="cuda")
    single_decode_with_kv_cache(q, k, v)
    files = list(tmp_path.glob("*.jsonl"))
    assert files
    for f in files:
        for line in f.read_text().splitlines():
            assert validate_trace_record(json.loads(line))
    importlib.reload(ft)
