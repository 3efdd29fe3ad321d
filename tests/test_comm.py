"""Multi-process comm tests on CPU (gloo, world_size=2) + Mapping unit tests.
The same code paths run over RCCL on the 8xMI355X node."""
import math
import os

import pytest
import torch
import torch.multiprocessing as mp

from flashinfer_amd.comm.mapping import Mapping


def test_mapping_tp_pp_cp():
    m = Mapping(world_size=8, rank=5, tp_size=2, pp_size=2, cp_size=2)
    # rank = pp*4 + cp*2 + tp: 5 = 1*4 + 0*2 + 1
    assert m.pp_rank == 1 and m.cp_rank == 0 and m.tp_rank == 1
    assert m.tp_group == [4, 5]
    assert m.cp_group == [5, 7]
    assert m.pp_group == [1, 5]
    assert m.is_last_pp_rank()
    assert m.pp_layers(9)[0] == 5  # 5 layers on pp0, 4 on pp1


def test_mapping_moe_ep():
    m = Mapping(world_size=8, rank=3, tp_size=8, moe_ep_size=4)
    assert m.moe_tp_size == 2 and m.moe_ep_size == 4
    assert m.moe_ep_rank == 3 and m.moe_tp_rank == 0
    assert m.moe_ep_group == [0, 1, 2, 3]
    assert m.ep_experts(8) == [6, 7]


def test_mapping_invalid():
    with pytest.raises(ValueError):
        Mapping(world_size=8, rank=0, tp_size=3)


def _worker(rank, world, port, fn_name):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)
    try:
        globals()[fn_name](rank, world)
    finally:
        dist.destroy_process_group()


def _run_mp(fn_name, world=2, port=29511):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, world, port, fn_name))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
    for p in procs:
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


# ---- worker bodies (module-level, spawn-picklable) ----

def _ulysses_body(rank, world):
    from flashinfer_amd.comm.ulysses import UlyssesCommunicator

    B, S_local, H, D = 2, 4, 8, 16
    torch.manual_seed(7)  # same full tensor on all ranks
    full = torch.randn(B, S_local * world, H, D)
    x = full[:, rank * S_local : (rank + 1) * S_local]
    comm = UlyssesCommunicator()
    y = comm.scatter_heads(x)
    # expect: full sequence, my head slice
    hl = H // world
    expected = full[:, :, rank * hl : (rank + 1) * hl]
    assert torch.equal(y, expected), "scatter_heads mismatch"
    back = comm.gather_heads(y)
    assert torch.equal(back, x), "gather_heads not inverse"


def _moe_a2a_body(rank, world):
    from flashinfer_amd.comm.moe_alltoall import MoeAlltoAll

    T, hidden, E, K = 5, 8, 4, 2
    torch.manual_seed(100 + rank)
    x = torch.randn(T, hidden)
    topk_ids = torch.randint(0, E, (T, K))
    topk_w = torch.softmax(torch.randn(T, K), -1)
    a2a = MoeAlltoAll(num_experts=E, top_k=K)
    recv_x, recv_local_exp, state = a2a.dispatch(x, topk_ids)
    # expert fn: multiply by (global_expert_id + 1)
    global_exp = recv_local_exp + rank * (E // world)
    expert_out = recv_x * (global_exp + 1).unsqueeze(1).float()
    out = a2a.combine(expert_out, topk_w, state)
    ref = torch.zeros_like(x)
    for t in range(T):
        for j in range(K):
            ref[t] += topk_w[t, j] * x[t] * (topk_ids[t, j] + 1).float()
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def _ar_fusion_body(rank, world):
    from flashinfer_amd.comm.allreduce import (
        AllReduceFusionPattern,
        allreduce_fusion,
    )

    torch.manual_seed(50 + rank)
    x = torch.randn(4, 64)
    torch.manual_seed(77)  # residual/weight shared
    res = torch.randn(4, 64)
    w = torch.randn(64)
    xs = [None] * world
    import torch.distributed as dist

    dist.all_gather_object(xs, x)
    x_sum = sum(xs)
    ref_res = res + x_sum
    rms = torch.rsqrt(ref_res.float().pow(2).mean(-1, keepdim=True) + 1e-6)
    ref_out = (ref_res.float() * rms * w).to(x.dtype)
    out, new_res = allreduce_fusion(
        x.clone(), res.clone(), w, 1e-6, AllReduceFusionPattern.kARResidualRMSNorm
    )
    assert torch.allclose(new_res, ref_res, atol=1e-5)
    assert torch.allclose(out, ref_out, atol=1e-4), (out - ref_out).abs().max()


def _ring_body(rank, world):
    from flashinfer_amd.parallel_attention import ring_attention

    M, L_local, Hq, Hkv, D = 6, 5, 4, 2, 16
    torch.manual_seed(9)  # same everywhere
    k_full = torch.randn(L_local * world, Hkv, D)
    v_full = torch.randn(L_local * world, Hkv, D)
    q_all = torch.randn(world, M, Hq, D)
    q = q_all[rank]
    # my kv shard
    k = k_full[rank * L_local : (rank + 1) * L_local]
    v = v_full[rank * L_local : (rank + 1) * L_local]
    out = ring_attention(q, k, v)
    # reference: full attention
    g = Hq // Hkv
    kf = k_full.float().repeat_interleave(g, dim=1)
    vf = v_full.float().repeat_interleave(g, dim=1)
    logits = torch.einsum("mhd,lhd->hml", q.float(), kf) / math.sqrt(D)
    p = torch.softmax(logits, -1)
    ref = torch.einsum("hml,lhd->mhd", p, vf)
    assert torch.allclose(out.float(), ref, atol=1e-4), (out.float() - ref).abs().max()


def test_ulysses_gloo():
    _run_mp("_ulysses_body", port=29512)


def test_moe_alltoall_gloo():
    _run_mp("_moe_a2a_body", port=29513)


def test_allreduce_fusion_gloo():
    _run_mp("_ar_fusion_body", port=29514)


def test_ring_attention_gloo():
    _run_mp("_ring_body", port=29515)


def test_eplb_rebalance():
    from flashinfer_amd.moe_ep import eplb_rebalance

    load = torch.tensor([100.0, 10, 10, 10, 10, 10, 10, 10])
    phy2log, replicas = eplb_rebalance(load, num_ranks=4, num_slots_per_rank=3)
    assert phy2log.shape == (4, 3)
    placed = phy2log[phy2log >= 0]
    # every expert placed at least once; the hottest gets the extra replicas
    assert set(placed.tolist()) == set(range(8))
    assert replicas.sum() == 12 and replicas[0] == replicas.max()
    # replica loads balance: max rank load close to mean
    rank_load = torch.zeros(4)
    for r in range(4):
        for s in range(3):
            e = int(phy2log[r, s])
            if e >= 0:
                rank_load[r] += float(load[e] / replicas[e])
    assert rank_load.max() <= rank_load.mean() * 1.6


def test_moe_ep_errors_and_enums():
    from flashinfer_amd import moe_ep

    assert moe_ep.supports_fault_tolerance()
    with pytest.raises(moe_ep.MoEEpTransportError):
        raise moe_ep.MoEEpTransportError("dispatch", 3, "timeout")
    assert moe_ep.EpAlgorithm("split") == moe_ep.EpAlgorithm.SPLIT


def _ep_layer_body(rank, world):
    import torch.distributed as dist

    from flashinfer_amd.moe_ep import MoeEp

    E, K, T, H = 4, 2, 6, 8
    ep = MoeEp(E, K, enable_fault_tolerance=True)
    torch.manual_seed(42)  # same logits everywhere
    logits = torch.randn(T, E)
    w, ids = ep.route(logits)
    x = torch.randn(T, H) + rank
    recv_x, local_exp, state = ep.dispatch(x, ids)
    assert (local_exp >= 0).all() and (local_exp < ep.experts_per_rank).all()
    y = recv_x * 2.0  # stand-in expert: f(x) = 2x regardless of expert
    out = ep.combine(y, w, state)
    torch.testing.assert_close(out, 2.0 * x, atol=1e-5, rtol=1e-5)

    # fault tolerance: mask the other rank; all routes go local
    ep.mask_rank(1 - rank)
    w2, ids2 = ep.route(logits)
    own = (ids2 // ep.experts_per_rank == rank)
    assert own.all()
    ep.clear_faults()
    assert bool(ep.alive_mask().all())


def test_moe_ep_layer_gloo():
    _run_mp("_ep_layer_body", world=2, port=29531)


def _agmm_body(rank, world):
    from flashinfer_amd.comm.all_gather_matmul import all_gather_matmul

    torch.manual_seed(10 + rank)
    m, K, N = 4, 16, 8
    inp = torch.randn(m, K)
    w_shared = torch.ones(K, N) * 0.5  # same on all ranks
    out = all_gather_matmul(inp, w_shared)
    # reference
    import torch.distributed as dist
    gathered = [torch.empty_like(inp) for _ in range(world)]
    dist.all_gather(gathered, inp)
    ref = torch.cat(gathered) @ w_shared
    torch.testing.assert_close(out, ref, atol=1e-5, rtol=1e-5)


def test_all_gather_matmul_gloo():
    _run_mp("_agmm_body", world=2, port=29541)


def _qar_body(rank, world):
    from flashinfer_amd.comm.quantized_allreduce import quantized_all_reduce

    torch.manual_seed(20 + rank)
    x = torch.randn(4, 256 * world)  # numel divides world * scale_group(256)
    out = quantized_all_reduce(x, scale_group=256)
    import torch.distributed as dist
    ref = x.clone()
    dist.all_reduce(ref)
    # fp8 payloads: ~2^-3 relative error bound per element group
    torch.testing.assert_close(out, ref, atol=0.25, rtol=0.1)


def test_quantized_all_reduce_gloo():
    _run_mp("_qar_body", world=2, port=29551)


@pytest.mark.gpu
def test_hip_ipc_two_process_one_gpu():
    """Real hipIpc handle exchange: two processes share one GPU buffer."""
    import torch.multiprocessing as mp2

    ctx = mp2.get_context("spawn")
    q01 = ctx.Queue()
    q10 = ctx.Queue()
    procs = [ctx.Process(target=_ipc_worker, args=(r, q01, q10))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
    for p in procs:
        assert p.exitcode == 0, f"ipc worker failed with {p.exitcode}"


def _ipc_worker(rank, q01, q10):
    import ctypes

    import torch

    from flashinfer_amd.comm.hip_ipc import (HipRTLibrary, hipIpcMemHandle_t)

    torch.cuda.init()
    rt = HipRTLibrary()
    rt.hipSetDevice(0)
    if rank == 0:
        ptr = rt.hipMalloc(4096)
        rt.hipMemset(ptr, 0x5A, 4096)
        rt.hipDeviceSynchronize()
        q01.put(rt.hipIpcGetMemHandle(ptr).bytes_())
        assert q10.get(timeout=60) == "ok"
        rt.hipFree(ptr)
    else:
        handle = hipIpcMemHandle_t.from_bytes(q01.get(timeout=60))
        ptr = rt.hipIpcOpenMemHandle(handle)
        host = (ctypes.c_byte * 4096)()
        # hipMemcpyDeviceToHost = 2
        rt._check(rt.lib.hipMemcpy(ctypes.byref(host), ctypes.c_void_p(ptr),
                                   4096, 2), "hipMemcpy")
        assert all((b & 0xFF) == 0x5A for b in bytes(host))
        rt.hipIpcCloseMemHandle(ptr)
        q10.put("ok")


@pytest.mark.gpu
def test_custom_allreduce_one_gpu_two_procs():
    """hipIpc one-shot AR kernel: 2 processes on one GPU, gloo handle
    exchange, xGMI-style direct peer reads."""
    import torch.multiprocessing as mp3

    ctx = mp3.get_context("spawn")
    port = 29561
    procs = [ctx.Process(target=_car_worker, args=(r, 2, port))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
    for p in procs:
        assert p.exitcode == 0, f"custom AR worker failed with {p.exitcode}"


def _car_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    import flashinfer_amd  # loads the extension
    from flashinfer_amd.comm.custom_ar import CustomAllReduce

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.cuda.set_device(0)
        ar = CustomAllReduce(max_bytes=1 << 20, spin_limit=1 << 24)
        torch.manual_seed(rank)
        x = torch.randn(64, 256, dtype=torch.bfloat16, device="cuda")
        y = ar.all_reduce(x)
        # reference via gloo on cpu
        ref = x.float().cpu()
        dist.all_reduce(ref)
        torch.testing.assert_close(y.float().cpu(), ref, atol=3e-2, rtol=3e-2)
        # repeat to exercise the double-buffer slots
        for it in range(3):
            x2 = torch.randn(64, 256, dtype=torch.bfloat16, device="cuda")
            y2 = ar.all_reduce(x2)
            ref2 = x2.float().cpu()
            dist.all_reduce(ref2)
            torch.testing.assert_close(y2.float().cpu(), ref2, atol=3e-2,
                                       rtol=3e-2)
        # fused AR + residual + rmsnorm
        res = torch.randn(64, 256, dtype=torch.bfloat16, device="cuda")
        res_ref = res.clone()
        w = torch.randn(256, dtype=torch.bfloat16, device="cuda")
        x3 = torch.randn(64, 256, dtype=torch.bfloat16, device="cuda")
        o = ar.all_reduce_rmsnorm(x3, res, w)
        sx = x3.float().cpu()
        dist.all_reduce(sx)
        rf = res_ref.float().cpu() + sx
        nf = rf / torch.sqrt(rf.pow(2).mean(-1, keepdim=True) + 1e-6)
        torch.testing.assert_close(o.float().cpu(), nf * w.float().cpu(),
                                   atol=5e-2, rtol=5e-2)
        torch.testing.assert_close(res.float().cpu(), rf, atol=3e-2, rtol=3e-2)
        ar.close()
    finally:
        dist.destroy_process_group()


def _mega_worker(rank, world, port, fail):
    try:
        import os

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch
        import torch.distributed as dist

        from flashinfer_amd.moe_ep import EpAlgorithm, MoeEp

        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.manual_seed(17)  # same on all ranks (replicated test tensors)
        E, k, T, H = 8, 2, 64, 32
        ep = MoeEp(E, k, algorithm=EpAlgorithm.MEGA)
        x = torch.randn(T, H) + rank  # different tokens per rank
        logits = torch.randn(T, E) * (rank + 1)
        # per-expert toy weights (replicated)
        we = torch.randn(E, H, H) / H**0.5

        def expert_fn(rx, rexp):
            out = torch.empty_like(rx)
            base = ep.rank * ep.experts_per_rank
            for e_local in range(ep.experts_per_rank):
                m = rexp == e_local
                if m.any():
                    out[m] = rx[m] @ we[base + e_local].t().to(rx.dtype)
            return out

        y = ep.forward_mega(x, logits, expert_fn, n_chunks=4)
        # reference: dense local computation of the same routing
        w, ids = ep.route(logits)
        ref = torch.zeros_like(x)
        for t in range(T):
            for j in range(k):
                e = int(ids[t, j])
                ref[t] += float(w[t, j]) * (x[t] @ we[e].t().to(x.dtype))
        torch.testing.assert_close(y, ref, atol=1e-4, rtol=1e-4)
        # chunked result == unchunked result
        y1 = ep.forward_mega(x, logits, expert_fn, n_chunks=1)
        torch.testing.assert_close(y, y1, atol=1e-5, rtol=1e-5)
        dist.destroy_process_group()
    except Exception:
        import traceback

        traceback.print_exc()
        fail.put(rank)


def test_moe_ep_mega_mode_gloo():
    """Mega-mode chunked dispatch/compute/combine pipeline matches the dense
    reference and the unchunked path (2-rank gloo)."""
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    fail = ctx.Queue()
    ps = [ctx.Process(target=_mega_worker, args=(r, 2, 29591, fail))
          for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(180)
    assert all(p.exitcode == 0 for p in ps)
    assert fail.empty()


def _dcp_body(rank, world):
    """DCP q-scatter / o-gather vs a single-rank reference: each rank holds
    partial attention states for the FULL batch over its KV shard; the
    LSE merge must equal full-KV attention."""
    from flashinfer_amd.comm.dcp import dcp_gather_o, dcp_scatter_q

    B, H, D, L = 2, 4, 16, 32
    g = torch.Generator().manual_seed(7)
    q_full = torch.randn(world * B, H, D, generator=g)
    k = torch.randn(L * world, H, D, generator=g)
    v = torch.randn(L * world, H, D, generator=g)
    # scatter: local batch shard -> full batch everywhere
    q_local = q_full[rank * B : (rank + 1) * B]
    q_all = dcp_scatter_q(q_local)
    assert torch.equal(q_all, q_full)
    # partial attention on this rank's KV shard (fp32 reference math)
    ks = k[rank * L : (rank + 1) * L]
    vs = v[rank * L : (rank + 1) * L]
    logits = torch.einsum("bhd,lhd->bhl", q_full.float(), ks.float())
    m = logits.max(-1, keepdim=True).values
    p = torch.exp(logits - m)
    o_part = torch.einsum("bhl,lhd->bhd", p, vs.float()) / p.sum(-1)[..., None]
    lse_part = (m[..., 0] + p.sum(-1).log()) / torch.tensor(2.0).log()
    sl = slice(rank * B, (rank + 1) * B)
    o, lse = dcp_gather_o(o_part, lse_part, sl)
    # reference: full-KV attention for this rank's batch rows
    logits_f = torch.einsum("bhd,lhd->bhl", q_full[sl].float(), k.float())
    ref = torch.softmax(logits_f, -1) @ v.float().permute(1, 0, 2).reshape(
        H, world * L, D)[0] if False else torch.einsum(
        "bhl,lhd->bhd", torch.softmax(logits_f, -1), v.float())
    torch.testing.assert_close(o, ref, atol=1e-4, rtol=1e-4)


def test_dcp_gloo():
    _run_mp("_dcp_body", port=29517)


def _backend_body(rank, world):
    from flashinfer_amd.comm.comm_backend import TorchDistBackend

    be = TorchDistBackend()
    assert be.rank == rank and be.world_size == world
    objs = be.allgather_object({"r": rank})
    assert [o["r"] for o in objs] == list(range(world))
    assert be.broadcast_object("x" if rank == 0 else None, src=0) == "x"
    be.barrier()


def test_comm_backend_gloo():
    _run_mp("_backend_body", port=29518)


def _ep_zero_token_body(rank, world):
    """Reference regression class (repro_ikr_zero_token_idle): an idle DP
    rank forwards a ZERO-TOKEN batch while peers have work. The rank must
    still issue its alltoallv (skipping would deadlock the peers), and
    the tokens routed TO its experts must come back reduced correctly."""
    from flashinfer_amd.moe_ep import MoeEp

    E, K, H = 4, 2, 8
    ep = MoeEp(E, K)
    T = 6 if rank == 0 else 0   # rank 1 idle this step
    logits = torch.full((T, E), -10.0)
    if T:
        logits[:, E // world:] = 10.0  # route everything to rank 1's experts
    w, ids = ep.route(logits)
    x = torch.randn(T, H)
    recv_x, local_exp, state = ep.dispatch(x, ids)
    if rank == 1:
        # the idle rank received rank 0's tokens for its experts
        assert recv_x.shape[0] == 6 * K
    y = recv_x * 2.0
    out = ep.combine(y, w, state)
    assert out.shape == (T, H)
    if T:
        torch.testing.assert_close(out, 2.0 * x, atol=1e-5, rtol=1e-5)


def test_moe_ep_zero_token_rank_gloo():
    _run_mp("_ep_zero_token_body", port=29519)
