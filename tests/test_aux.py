"""CPU tests: aux subsystems — logging, trace, autotuner, CLI."""
import json
import os
import subprocess
import sys

import pytest
import torch


def test_cli_show_config_and_module_status():
    env = dict(os.environ)
    r = subprocess.run([sys.executable, "-m", "flashinfer_amd", "show-config"],
                       capture_output=True, text=True, env=env)
    assert r.returncode == 0 and "flashinfer_amd" in r.stdout
    r = subprocess.run([sys.executable, "-m", "flashinfer_amd", "module-status"],
                       capture_output=True, text=True, env=env)
    assert "native ops" in r.stdout or "NOT BUILT" in r.stdout


def test_api_logging_env(tmp_path):
    log = tmp_path / "api.log"
    env = dict(os.environ, FLASHINFER_LOGLEVEL="3", FLASHINFER_LOGDEST=str(log))
    code = (
        "import torch\n"
        "from flashinfer_amd.comm.mapping import Mapping\n"
        "from flashinfer_amd.api_logging import flashinfer_api\n"
        "@flashinfer_api\n"
        "def f(x): return x\n"
        "f(torch.zeros(2, 3))\n"
    )
    r = subprocess.run([sys.executable, "-c", code], capture_output=True, text=True,
                       env=env)
    assert r.returncode == 0, r.stderr
    assert "f(" in log.read_text()
    assert "Tensor(2, 3)" in log.read_text()


def test_fi_trace_dump(tmp_path):
    env = dict(os.environ, FLASHINFER_TRACE_DUMP=str(tmp_path))
    code = (
        "import torch\n"
        "from flashinfer_amd.fi_trace import fi_trace\n"
        "@fi_trace\n"
        "def myop(x, k=3): return x\n"
        "myop(torch.zeros(4, 8), k=5)\n"
        "myop(torch.zeros(4, 8), k=5)\n"  # dedup
    )
    r = subprocess.run([sys.executable, "-c", code], capture_output=True, text=True,
                       env=env)
    assert r.returncode == 0, r.stderr
    lines = (tmp_path / "myop.jsonl").read_text().strip().splitlines()
    assert len(lines) == 1
    rec = json.loads(lines[0])
    assert rec["args"][0]["shape"] == [4, 8] and rec["kwargs"]["k"] == 5


def test_autotuner_cache_selection():
    from flashinfer_amd import autotuner

    calls = []
    r = autotuner.TunableRunner(
        "dummy", [lambda x: calls.append(0) or x, lambda x: calls.append(1) or x],
        key_fn=lambda x: (len(x),),
    )
    out = r.run([1, 2, 3])  # not tuning: tactic 0
    assert out == [1, 2, 3] and calls == [0]


@pytest.mark.gpu
def test_intra_kernel_profiler():
    """Prefill with a profiler buffer emits decodable begin/end event pairs."""
    import flashinfer_amd as fi
    from flashinfer_amd import profiler

    torch.manual_seed(0)
    q = torch.randn(256, 8, 128, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(512, 2, 128, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(512, 2, 128, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(16 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    qo = torch.tensor([0, 256], dtype=torch.int32, device="cuda")
    kvi = torch.tensor([0, 512], dtype=torch.int32, device="cuda")
    w.plan(qo, kvi, 8, 2, 128, causal=True)
    buf = profiler.make_profiler_buffer(1 << 14)
    out = w.run(q, k, v, profiler_buffer=buf)
    assert out.isfinite().all()
    events = profiler.decode_events(buf)
    assert events, "no profiler events recorded"
    # every block must emit tile begin (0) and end, kv_mainloop begin/end
    by_block = {}
    for blk, eidx, etype, ts in events:
        by_block.setdefault(blk, []).append((eidx, etype))
    for blk, evs in by_block.items():
        assert (0, 0) in evs and (0, 1) in evs and (1, 0) in evs and (1, 1) in evs
    # run without buffer still fine (null-pointer fast path)
    out2 = w.run(q, k, v)
    assert torch.equal(out, out2)
    import tempfile, os, json
    fn = os.path.join(tempfile.mkdtemp(), "trace.json")
    profiler.export_to_chrome_trace(buf, ["tile", "kv_mainloop"], fn)
    tr = json.load(open(fn))
    assert len(tr["traceEvents"]) >= 4


@pytest.mark.gpu
def test_pod_wrapper_concurrent_streams():
    """POD outputs must match the same prefill/decode run separately."""
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, page = 8, 2, 128, 16
    q_p = torch.randn(256, Hq, D, dtype=torch.bfloat16, device="cuda")
    k_p = torch.randn(256, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v_p = torch.randn(256, Hkv, D, dtype=torch.bfloat16, device="cuda")
    batch, pages_per = 8, 16
    npages = batch * pages_per
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    indptr = torch.arange(0, npages + 1, pages_per, dtype=torch.int32, device="cuda")
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((batch,), page, dtype=torch.int32, device="cuda")
    q_d = torch.randn(batch, Hq, D, dtype=torch.bfloat16, device="cuda")

    ws = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    pod = fi.PODWithPagedKVCacheWrapper(ws, "NHD")
    pod.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    o_p, o_d = pod.run(q_p, k_p, v_p, q_d, (kc, vc))
    torch.cuda.synchronize()

    ref_p = fi.single_prefill_with_kv_cache(q_p, k_p, v_p, causal=True)
    ws2 = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    dec = fi.BatchDecodeWithPagedKVCacheWrapper(ws2, "NHD")
    dec.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    ref_d = dec.run(q_d, (kc, vc))
    torch.testing.assert_close(o_p.float(), ref_p.float())
    torch.testing.assert_close(o_d.float(), ref_d.float())


@pytest.mark.gpu
def test_batch_pod_wrapper():
    """BatchPOD prefill+decode on two streams match separate runs."""
    import flashinfer_amd as fi

    torch.manual_seed(3)
    Hq, Hkv, D, page = 8, 2, 128, 16
    # prefill batch: 2 reqs x 64 q tokens, kv = 128 each
    bp, qo_len, kvlen = 2, 64, 128
    pp = kvlen // page
    np_p = bp * pp
    kcp = torch.randn(np_p, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vcp = torch.randn(np_p, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    qo_indptr_p = torch.arange(0, (bp + 1) * qo_len, qo_len,
                               dtype=torch.int32, device="cuda")
    kv_indptr_p = torch.arange(0, np_p + 1, pp, dtype=torch.int32, device="cuda")
    kv_indices_p = torch.arange(np_p, dtype=torch.int32, device="cuda")
    lpl_p = torch.full((bp,), page, dtype=torch.int32, device="cuda")
    q_p = torch.randn(bp * qo_len, Hq, D, dtype=torch.bfloat16, device="cuda")
    # decode batch
    bd, ppd = 4, 8
    np_d = bd * ppd
    kcd = torch.randn(np_d, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vcd = torch.randn(np_d, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    qo_indptr_d = torch.arange(0, bd + 1, dtype=torch.int32, device="cuda")
    kv_indptr_d = torch.arange(0, np_d + 1, ppd, dtype=torch.int32, device="cuda")
    kv_indices_d = torch.arange(np_d, dtype=torch.int32, device="cuda")
    lpl_d = torch.full((bd,), page, dtype=torch.int32, device="cuda")
    q_d = torch.randn(bd, Hq, D, dtype=torch.bfloat16, device="cuda")

    wsp = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    wsd = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    pod = fi.BatchPODWithPagedKVCacheWrapper(wsp, wsd)
    pod.plan(qo_indptr_p, kv_indptr_p, kv_indices_p, lpl_p,
             qo_indptr_d, kv_indptr_d, kv_indices_d, lpl_d,
             Hq, Hkv, D, page, q_data_type=torch.bfloat16, causal_p=True)
    o_p, o_d = pod.run(q_p, (kcp, vcp), q_d, (kcd, vcd))
    torch.cuda.synchronize()

    ws2 = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    pf = fi.BatchPrefillWithPagedKVCacheWrapper(ws2, "NHD")
    pf.plan(qo_indptr_p, kv_indptr_p, kv_indices_p, lpl_p, Hq, Hkv, D, page,
            causal=True)
    ref_p = pf.run(q_p, (kcp, vcp))
    ws3 = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    dec = fi.BatchDecodeWithPagedKVCacheWrapper(ws3, "NHD")
    dec.plan(kv_indptr_d, kv_indices_d, lpl_d, Hq, Hkv, D, page,
             q_data_type=torch.bfloat16)
    ref_d = dec.run(q_d, (kcd, vcd))
    torch.testing.assert_close(o_p.float(), ref_p.float())
    torch.testing.assert_close(o_d.float(), ref_d.float())


@pytest.mark.gpu
def test_testing_bench_utils():
    from flashinfer_amd.testing import (
        attention_tflops_per_sec_with_actual_seq_lens, bench_gpu_time)

    x = torch.randn(1024, 1024, device="cuda", dtype=torch.bfloat16)
    times = bench_gpu_time(lambda: torch.mm(x, x), dry_run_iters=2,
                           repeat_iters=5)
    assert len(times) == 5 and all(t > 0 for t in times)
    tf = attention_tflops_per_sec_with_actual_seq_lens(
        torch.tensor([128]), torch.tensor([1024]), 128, 128, 32, True, 1.0)
    assert tf > 0


def test_trace_apply_substitution():
    import flashinfer_amd as fi
    from flashinfer_amd import trace_apply

    calls = []

    def fake_rmsnorm(input, weight, *a, **kw):
        calls.append(input.shape)
        return input * 0 + 7

    try:
        trace_apply.enable_apply({"rmsnorm": fake_rmsnorm})
        assert trace_apply.is_enabled()
        x = torch.randn(2, 8)
        w = torch.randn(8)
        out = fi.rmsnorm(x, w)
        assert calls and (out == 7).all()
        st = trace_apply.stats()
        assert st[("rmsnorm", "hit")] >= 1
    finally:
        trace_apply.disable_apply()
    assert not trace_apply.is_enabled()


def test_collect_env_runs():
    from flashinfer_amd.collect_env import collect_env

    info = collect_env()
    assert "torch" in info and "extension_loaded" in info
