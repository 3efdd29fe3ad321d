"""LogitsPipe: compile-time fusion (CPU) + GPU execution vs unfused ops."""
import pytest
import torch

from flashinfer_amd.logits_processor import (
    LegalizationError, LogitsPipe, MinP, Sample, Softmax, Temperature,
    TensorType, TopK, TopP, compile_pipeline,
)


def test_pipeline_fusion_plan():
    pipe = LogitsPipe([Temperature(), Softmax(), TopK(), TopP(), Sample()])
    names = [getattr(op, "name", type(op).__name__) for op in pipe.compiled_ops]
    assert names == ["temperature_softmax", "top_k_top_p_sample"]
    pipe2 = LogitsPipe([TopK(), Sample()], input_type=TensorType.PROBS)
    assert [op.name for op in pipe2.compiled_ops] == ["top_k_sample"]
    # TopK on logits cannot fuse with Sample's probs kernel
    pipe3 = LogitsPipe([TopK(), Sample()], input_type=TensorType.LOGITS)
    assert [type(op).__name__ for op in pipe3.compiled_ops] == ["TopK", "Sample"]


def test_pipeline_legalization_errors():
    with pytest.raises(LegalizationError):
        compile_pipeline([Sample(), TopK()], TensorType.PROBS)  # Sample not last
    with pytest.raises(LegalizationError):
        compile_pipeline([TopP()], TensorType.LOGITS)  # TopP needs probs
    with pytest.raises(ValueError):
        LogitsPipe([])


@pytest.mark.gpu
def test_pipeline_executes_and_matches_unfused():
    torch.manual_seed(0)
    logits = torch.randn(16, 32000, device="cuda")
    pipe = LogitsPipe([Temperature(), Softmax(), TopK(), TopP(), Sample()])
    gen = torch.Generator(device="cuda").manual_seed(7)
    ids = pipe(logits, temperature=0.8, top_k=40, top_p=0.9, generator=gen)
    assert ids.shape == (16,) and (ids >= 0).all() and (ids < 32000).all()

    import flashinfer_amd as fi
    gen2 = torch.Generator(device="cuda").manual_seed(7)
    probs = fi.softmax(logits, temperature=0.8)
    ref = fi.top_k_top_p_sampling_from_probs(probs, 40, 0.9, generator=gen2)
    assert torch.equal(ids, ref)


@pytest.mark.gpu
def test_pipeline_probs_input_and_minp():
    torch.manual_seed(1)
    probs = torch.softmax(torch.randn(8, 1000, device="cuda"), -1)
    pipe = LogitsPipe([MinP(), Sample()], input_type=TensorType.PROBS)
    assert [op.name for op in pipe.compiled_ops] == ["min_p_sample"]
    ids = pipe(probs, min_p=0.05)
    assert ids.shape == (8,)
    # renorm-only pipeline keeps PROBS output
    pipe2 = LogitsPipe([TopK(), TopP()], input_type=TensorType.PROBS)
    out = pipe2(probs, top_k=50, top_p=0.9)
    torch.testing.assert_close(out.sum(-1), torch.ones(8, device="cuda"),
                               atol=1e-4, rtol=1e-4)
