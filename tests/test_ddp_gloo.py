"""Multi-process data-parallel correctness on CPU (gloo, world_size=2).

The reference validates its distributed path single-machine via Spark
local[4] (Java:318); here the RCCL path is validated with the gloo backend
— same torch.distributed code path, no GPU needed.
"""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, fn_name, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29531"
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        fn = globals()[fn_name]
        result = fn(rank, world)
        # serialize tensors as plain lists: shared-memory fds die with the child
        def plain(v):
            if isinstance(v, torch.Tensor):
                return v.detach().cpu().tolist()
            if isinstance(v, (list, tuple)):
                return [plain(x) for x in v]
            return v

        q.put((rank, plain(result)))
    finally:
        dist.destroy_process_group()


def _run_mp(fn_name, world=2, tmpdir=""):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(r, world, fn_name, tmpdir, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    # drain BEFORE join: large results (full param vectors) overflow the
    # pipe buffer, and a child blocked on q.put never exits
    out = {}
    for _ in procs:
        r, v = q.get(timeout=240)
        out[r] = v
    for p in procs:
        p.join(60)
        assert p.exitcode == 0, f"worker failed: exit {p.exitcode}"
    return out


# ------------------------------------------------------------------ bodies
def _body_grad_reducer(rank, world):
    from gan_deeplearning4j_amd.parallel.ddp import GradReducer

    torch.manual_seed(0)  # same init on all ranks
    m = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 1))
    red = GradReducer([m], bucket_cap_mb=1)
    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(4, 8)
    red.prepare()
    m(x).sum().backward()
    red.finish()
    return [p.grad.clone() for p in m.parameters()]


def _body_trainer_sync(rank, world):
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.models import build_mlp_gan
    from gan_deeplearning4j_amd.train import GanTrainer

    cfg = preset("mlp_tabular_cpu")
    cfg.train.use_gpu = False
    gen, dis = build_mlp_gan(cfg, hidden=16)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    torch.manual_seed(500 + rank)
    for _ in range(2):
        tr.step(torch.rand(8, cfg.data.num_features))
    # replicas must stay bit-identical after synced updates
    return tr.dis.params_flat()


def _body_trainer_sync_conv(rank, world):
    """Conv-model (dcgan28) replicas stay synced — the same path the
    driver's multi-GPU scale bench exercises, minus RCCL."""
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.models import build_dcgan
    from gan_deeplearning4j_amd.train import GanTrainer

    cfg = preset("dcgan28")
    cfg.train.use_gpu = False
    gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    torch.manual_seed(700 + rank)  # different data per rank
    for _ in range(2):
        real = torch.rand(4, 1, 28, 28) * 2 - 1
        tr.step(real)
    return [tr.dis.params_flat(), tr.gen.params_flat()]


def _body_param_averaging(rank, world):
    from gan_deeplearning4j_amd.parallel.ddp import average_parameters

    m = torch.nn.Sequential(torch.nn.Linear(4, 4, bias=False),
                            torch.nn.BatchNorm1d(4))
    with torch.no_grad():
        m[0].weight.fill_(float(rank + 1))
        m[1].running_mean.fill_(float(rank))  # 0.0 / 1.0 -> 0.5
    average_parameters(m)
    return [m[0].weight.detach().clone(),
            m[1].running_mean.detach().clone()]


def _body_grad_reducer_exact(rank, world):
    """Reduced grads must equal the mean of the per-rank local grads."""
    from gan_deeplearning4j_amd.parallel.ddp import GradReducer

    torch.manual_seed(0)
    m = torch.nn.Linear(8, 4)
    torch.manual_seed(100 + rank)
    x = torch.randn(4, 8)
    # local gradient, no reducer
    m(x).sum().backward()
    local = [p.grad.clone() for p in m.parameters()]
    for p in m.parameters():
        p.grad = None
    red = GradReducer([m], bucket_cap_mb=1)
    red.prepare()
    m(x).sum().backward()
    red.finish()
    return [local, [p.grad.clone() for p in m.parameters()]]


def _body_local_steps(rank, world):
    """local_steps=2 (the averaging_frequency analog): step 1 keeps grads
    local, step 2 reduces."""
    from gan_deeplearning4j_amd.parallel.ddp import GradReducer

    torch.manual_seed(0)
    m = torch.nn.Linear(4, 1, bias=False)
    red = GradReducer([m], bucket_cap_mb=1, local_steps=2)
    torch.manual_seed(100 + rank)
    x = torch.randn(2, 4)
    red.prepare()
    m(x).sum().backward()
    red.finish()
    g1 = m.weight.grad.clone()
    m.weight.grad = None
    red.prepare()
    m(x).sum().backward()
    red.finish()
    return [g1, m.weight.grad.clone()]


def _body_broadcast(rank, world):
    from gan_deeplearning4j_amd.parallel.ddp import broadcast_parameters

    m = torch.nn.Linear(4, 1, bias=False)
    with torch.no_grad():
        m.weight.fill_(float(rank + 1))
    broadcast_parameters(m, src=0)
    return m.weight.detach().clone()


# ------------------------------------------------------------------- tests
def test_grad_reducer_averages_across_ranks():
    out = _run_mp("_body_grad_reducer")
    for g0, g1 in zip(out[0], out[1]):
        assert torch.allclose(torch.tensor(g0), torch.tensor(g1), atol=1e-6)


def test_trainer_replicas_stay_synced():
    out = _run_mp("_body_trainer_sync")
    assert torch.allclose(torch.tensor(out[0]), torch.tensor(out[1]), atol=1e-5)


def test_conv_trainer_replicas_stay_synced():
    out = _run_mp("_body_trainer_sync_conv")
    for i in range(2):
        a, b = torch.tensor(out[0][i]), torch.tensor(out[1][i])
        assert torch.allclose(a, b, atol=1e-5)


def test_parameter_averaging():
    out = _run_mp("_body_param_averaging")
    # ranks filled 1.0 and 2.0 -> average 1.5 on both; BN running stats
    # (buffers) average too (0.0/1.0 -> 0.5)
    for r in (0, 1):
        w, mean = (torch.tensor(t) for t in out[r])
        assert torch.allclose(w, torch.full((4, 4), 1.5))
        assert torch.allclose(mean, torch.full((4,), 0.5))


def test_grad_reducer_exact_mean():
    out = _run_mp("_body_grad_reducer_exact")
    local0, _ = out[0]
    local1, red0 = out[1][0], out[0][1]
    red1 = out[1][1]
    for l0, l1, r0, r1 in zip(local0, local1, red0, red1):
        want = (torch.tensor(l0) + torch.tensor(l1)) / 2
        assert torch.allclose(torch.tensor(r0), want, atol=1e-6)
        assert torch.allclose(torch.tensor(r1), want, atol=1e-6)


def test_local_steps_gates_reduction():
    out = _run_mp("_body_local_steps")
    g1_r0, g2_r0 = (torch.tensor(t) for t in out[0])
    g1_r1, g2_r1 = (torch.tensor(t) for t in out[1])
    # step 1 (local): per-rank data differs -> grads differ
    assert not torch.allclose(g1_r0, g1_r1, atol=1e-6)
    # step 2 (reduced): grads identical and equal to the mean of the locals
    assert torch.allclose(g2_r0, g2_r1, atol=1e-6)
    assert torch.allclose(g2_r0, (g1_r0 + g1_r1) / 2, atol=1e-6)


def test_broadcast_parameters():
    out = _run_mp("_body_broadcast")
    assert torch.allclose(torch.tensor(out[0]), torch.full((1, 4), 1.0))
    assert torch.allclose(torch.tensor(out[1]), torch.full((1, 4), 1.0))


def _body_comm_dtype_env(rank, world):
    """gloo never uses bf16 comms even when requested (fp32 fallback)."""
    import os

    from gan_deeplearning4j_amd.parallel.ddp import GradReducer

    os.environ["GDLJ_COMM_DTYPE"] = "bf16"
    try:
        m = torch.nn.Linear(4, 2).to(torch.bfloat16)
        red = GradReducer([m], bucket_cap_mb=1)
        return [str(red.comm_dtype)]
    finally:
        del os.environ["GDLJ_COMM_DTYPE"]


def test_comm_dtype_gloo_fp32_fallback():
    out = _run_mp("_body_comm_dtype_env")
    assert out[0][0] == "torch.float32"
    assert out[1][0] == "torch.float32"
