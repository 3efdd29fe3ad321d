"""CPU tests: aux subsystems — logging, trace, autotuner, CLI."""
import json
import os
import subprocess
import sys


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
