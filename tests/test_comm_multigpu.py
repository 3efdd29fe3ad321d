"""Multi-GPU RCCL + hipIpc comm tests: run automatically on any box with >=2
GPUs (reference tests/comm/test_trtllm_allreduce.py:304 mp-spawn pattern);
skipped on single-GPU boxes."""
import os
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

_NEED = pytest.mark.skipif(
    not torch.cuda.is_available() or torch.cuda.device_count() < 2,
    reason="needs >= 2 GPUs")


def _ar_worker(rank, world, port, fail):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        import torch
        import torch.distributed as dist

        from flashinfer_amd.comm.custom_ar import CustomAllReduce

        dist.init_process_group("nccl", rank=rank, world_size=world)
        torch.cuda.set_device(rank)
        ar = CustomAllReduce(max_bytes=8 << 20)
        torch.manual_seed(100 + rank)
        for numel in (4096, 1 << 20):
            x = torch.randn(numel, dtype=torch.bfloat16, device="cuda")
            ref = x.clone()
            dist.all_reduce(ref)  # RCCL reference
            for strat in ("one_shot", "two_shot"):
                y = ar.all_reduce(x, strategy=strat)
                torch.cuda.synchronize()
                torch.testing.assert_close(y.float(), ref.float(),
                                           atol=5e-2, rtol=5e-2)
        # fused AR + rmsnorm vs composed reference
        d = 4096
        x = torch.randn(8, d, dtype=torch.bfloat16, device="cuda")
        res = torch.randn(8, d, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(d, dtype=torch.bfloat16, device="cuda")
        ref_sum = x.clone()
        dist.all_reduce(ref_sum)
        ref_res = (res.float() + ref_sum.float())
        rms = ref_res * torch.rsqrt(
            ref_res.pow(2).mean(-1, keepdim=True) + 1e-6)
        ref_out = (rms * w.float())
        res2 = res.clone()
        out = ar.all_reduce_rmsnorm(x, res2, w)
        torch.cuda.synchronize()
        torch.testing.assert_close(out.float(), ref_out, atol=7e-2, rtol=7e-2)
        ar.close()
        dist.destroy_process_group()
    except Exception:
        import traceback
        traceback.print_exc()
        fail.put(rank)


@_NEED
def test_custom_ar_2gpu_rccl():
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    fail = ctx.Queue()
    ps = [ctx.Process(target=_ar_worker, args=(r, 2, 29581, fail))
          for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(240)
    assert all(p.exitcode == 0 for p in ps)
    assert fail.empty()
