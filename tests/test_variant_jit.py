"""GPU: JIT attention-variant customization — user device code compiled into
the prefill kernel (reference csrc/batch_prefill_customize_config.jinja +
variants.cuh mechanism)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_custom_tanh_cap_variant():
    import flashinfer_amd as fi
    from flashinfer_amd.jit.attention import gen_customize_batch_prefill_module

    mod = gen_customize_batch_prefill_module(
        "test_tanh_cap25", logits_transform="return 25.f * tanhf(s / 25.f);")
    torch.manual_seed(0)
    qo, kv, Hq, Hkv, D = 333, 333, 8, 2, 128
    q = torch.randn(qo, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD", jit_module=mod)
    qo_indptr = torch.tensor([0, qo], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0, kv], dtype=torch.int32, device="cuda")
    w.plan(qo_indptr, kv_indptr, Hq, Hkv, D, causal=True,
           q_data_type=torch.bfloat16)
    out = w.run(q, k, v)
    # reference: tanh-capped causal attention
    g = Hq // Hkv
    qf = q.float().transpose(0, 1)
    kf = k.float().repeat_interleave(g, 1).transpose(0, 1)
    vf = v.float().repeat_interleave(g, 1).transpose(0, 1)
    logits = qf @ kf.transpose(-1, -2) / math.sqrt(D)
    logits = 25.0 * torch.tanh(logits / 25.0)
    qpos = torch.arange(qo, device="cuda")[:, None]
    kpos = torch.arange(kv, device="cuda")[None, :]
    logits = logits.masked_fill((kpos > qpos)[None], float("-inf"))
    ref = (torch.softmax(logits, -1) @ vf).transpose(0, 1)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_custom_mask_variant_banded():
    """Custom logits_mask: banded attention defined in user code."""
    import flashinfer_amd as fi
    from flashinfer_amd.jit.attention import gen_customize_batch_prefill_module

    mod = gen_customize_batch_prefill_module(
        "test_band64",
        logits_mask="return kv_idx >= (int64_t)qo_idx - 64 && "
                    "kv_idx <= (int64_t)qo_idx + 64;")
    torch.manual_seed(1)
    qo = kv = 256
    Hq, Hkv, D = 4, 4, 128
    q = torch.randn(qo, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD", jit_module=mod)
    qo_indptr = torch.tensor([0, qo], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0, kv], dtype=torch.int32, device="cuda")
    w.plan(qo_indptr, kv_indptr, Hq, Hkv, D, causal=False,
           q_data_type=torch.bfloat16)
    out = w.run(q, k, v)
    qf = q.float().transpose(0, 1)
    kf = k.float().transpose(0, 1)
    vf = v.float().transpose(0, 1)
    logits = qf @ kf.transpose(-1, -2) / math.sqrt(D)
    qpos = torch.arange(qo, device="cuda")[:, None]
    kpos = torch.arange(kv, device="cuda")[None, :]
    band = (kpos >= qpos - 64) & (kpos <= qpos + 64)
    logits = logits.masked_fill(~band[None], float("-inf"))
    ref = (torch.softmax(logits, -1) @ vf).transpose(0, 1)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
