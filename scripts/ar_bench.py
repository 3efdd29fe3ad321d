"""2-process/1-GPU one-shot allreduce latency probe (same-die stand-in for
the 8-GPU xGMI deployment; the kernel path is identical)."""
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch
    import torch.distributed as dist

    import flashinfer_amd  # noqa: F401
    from flashinfer_amd.comm.custom_ar import CustomAllReduce

    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    ar = CustomAllReduce(max_bytes=32 << 20, spin_limit=1 << 26)
    for numel in (4096, 65536, 1 << 20, 4 << 20, 16 << 20):
        torch.manual_seed(7 + rank)
        x = torch.randn(numel, dtype=torch.bfloat16, device="cuda")
        ref_c = x.float().cpu()
        dist.all_reduce(ref_c)  # gloo CPU reference
        for strat in ("one_shot", "two_shot"):
            y = ar.all_reduce(x, strategy=strat)
            torch.cuda.synchronize()
            err = (y.float().cpu() - ref_c).abs().max().item()
            assert err < 0.1, f"{strat} numel={numel} err={err}"
            for _ in range(5):
                ar.all_reduce(x, strategy=strat)
            torch.cuda.synchronize()
            dist.barrier()
            t0 = time.perf_counter()
            for _ in range(20):
                ar.all_reduce(x, strategy=strat)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 20
            if rank == 0:
                print(f"{strat} AR world={world} {numel*2/1024:.0f} KiB: "
                      f"{dt*1e6:.1f} us  {numel*2/dt/1e9:.1f} GB/s")
    ar.close()
    dist.destroy_process_group()


if __name__ == "__main__":
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    ps = [ctx.Process(target=worker, args=(r, 2, 29571)) for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(180)
    sys.exit(max(p.exitcode for p in ps))
