"""Multi-process CPU (gloo) tests for the data-parallel layer."""

import multiprocessing as mp
import os

import pytest
import torch


def _worker(rank, world_size, port, q, delay):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from noisynet_amd import distributed as dist_mod

    dist_mod.init_distributed(backend="gloo")
    torch.manual_seed(1234)  # same init on all ranks

    model = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
    dp = dist_mod.DataParallel(model, bucket_cap_mb=1, delay_allreduce=delay)

    # rank-dependent data
    torch.manual_seed(100 + rank)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    dp.finish()

    grad = model[0].weight.grad.clone()
    # reference: average of per-rank grads computed serially
    ref_grads = []
    for r in range(world_size):
        m2 = torch.nn.Sequential(
            torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
        torch.manual_seed(1234)
        for p_src, p_dst in zip(model.parameters(), m2.parameters()):
            pass
        m2.load_state_dict({k: v.detach().clone() for k, v in model.state_dict().items()})
        torch.manual_seed(100 + r)
        xr = torch.randn(8, 16)
        yr = torch.randn(8, 4)
        lr = torch.nn.functional.mse_loss(m2(xr), yr)
        lr.backward()
        ref_grads.append(m2[0].weight.grad)
    expected = torch.stack(ref_grads).mean(0)

    ok = torch.allclose(grad, expected, atol=1e-6)
    q.put((rank, bool(ok)))
    dist.destroy_process_group()


@pytest.mark.parametrize("delay", [False, True])
def test_dataparallel_grad_allreduce_gloo(delay):
    world_size = 2
    port = 29511 + (1 if delay else 0)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world_size, port, q, delay))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world_size)]
    for p in procs:
        p.join(timeout=120)
    assert all(ok for _, ok in results), results


def _bench_worker(rank, world_size, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import subprocess
    import sys
    # run bench.py in-process style: import main
    sys.argv = ["bench.py", "--steps", "2", "--warmup", "1", "--batch", "16"]
    import importlib
    import io
    from contextlib import redirect_stdout
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, repo)
    bench = importlib.import_module("bench")
    buf = io.StringIO()
    with redirect_stdout(buf):
        bench.main()
    q.put((rank, buf.getvalue()))


@pytest.mark.parametrize("world_size", [2, 8])
def test_bench_multiprocess_cpu(world_size):
    """bench.py runs under the torchrun env contract with gloo -- ws=2 and
    the full 8-rank shape the driver's SCALE run will launch (dry-run of
    rendezvous, bucket all-reduce and teardown at dp8)."""
    import json
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_bench_worker,
                         args=(r, world_size, 29521 + world_size, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    outs = {r: o for r, o in (q.get(timeout=600) for _ in range(world_size))}
    for p in procs:
        p.join(timeout=300)
    # rank 0 printed exactly one JSON line
    lines = [l for l in outs[0].strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == world_size
    assert rec["scaling"] == "weak"
    assert rec["value"] > 0
    for r in range(1, world_size):
        assert "{" not in outs[r]


def _missing_grad_worker(rank, world_size, port, q, delay):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from noisynet_amd import distributed as dist_mod

    dist_mod.init_distributed(backend="gloo")
    torch.manual_seed(55)

    class M(torch.nn.Module):
        """Model with manually-updated clip params (w_max1-style,
        drivers/cifar.py:283) that never receive autograd grads, sharing
        buckets with real params; plus an unused layer."""

        def __init__(self):
            super().__init__()
            self.fc1 = torch.nn.Linear(16, 32)
            self.w_max1 = torch.nn.Parameter(torch.tensor(0.25))
            self.fc2 = torch.nn.Linear(32, 4)
            self.unused = torch.nn.Linear(8, 8)
            self.w_min1 = torch.nn.Parameter(torch.tensor(-0.25))

        def forward(self, x):
            return self.fc2(torch.relu(self.fc1(x)))

    model = M()
    dp = dist_mod.DataParallel(model, bucket_cap_mb=1, delay_allreduce=delay)
    torch.manual_seed(300 + rank)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    torch.nn.functional.mse_loss(model(x), y).backward()
    dp.finish()  # must not hang or skip fc* grads

    grad = model.fc1.weight.grad.clone()
    ref = []
    for r in range(world_size):
        torch.manual_seed(55)
        m2 = M()
        torch.manual_seed(300 + r)
        xr, yr = torch.randn(8, 16), torch.randn(8, 4)
        torch.nn.functional.mse_loss(m2(xr), yr).backward()
        ref.append(m2.fc1.weight.grad)
    expected = torch.stack(ref).mean(0)
    ok = torch.allclose(grad, expected, atol=1e-6)
    ok_none = model.w_max1.grad is None and model.unused.weight.grad is None
    q.put((rank, bool(ok and ok_none)))
    dist.destroy_process_group()


@pytest.mark.parametrize("world_size,delay", [(2, False), (4, False), (4, True)])
def test_dataparallel_missing_grads_and_ws4(world_size, delay):
    """Buckets containing never-gradded params (manual w_max updates) and
    unused layers still reduce the real grads identically on every rank --
    at world_size 4 as well as 2 (8-GPU readiness drill)."""
    port = 29541 + world_size + (100 if delay else 0)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_missing_grad_worker,
                         args=(r, world_size, port, q, delay))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world_size)]
    for p in procs:
        p.join(timeout=120)
    assert all(ok for _, ok in results), results


def _accum_worker(rank, world_size, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from noisynet_amd import distributed as dist_mod

    dist_mod.init_distributed(backend="gloo")
    torch.manual_seed(66)
    model = torch.nn.Linear(16, 4)
    dp = dist_mod.DataParallel(model, bucket_cap_mb=1)
    torch.manual_seed(400 + rank)
    x1, y1 = torch.randn(8, 16), torch.randn(8, 4)
    x2, y2 = torch.randn(8, 16), torch.randn(8, 4)
    with dp.no_sync():
        torch.nn.functional.mse_loss(model(x1), y1).backward()
    torch.nn.functional.mse_loss(model(x2), y2).backward()
    dp.finish()
    grad = model.weight.grad.clone()

    ref = []
    for r in range(world_size):
        torch.manual_seed(66)
        m2 = torch.nn.Linear(16, 4)
        torch.manual_seed(400 + r)
        a1, b1 = torch.randn(8, 16), torch.randn(8, 4)
        a2, b2 = torch.randn(8, 16), torch.randn(8, 4)
        torch.nn.functional.mse_loss(m2(a1), b1).backward()
        torch.nn.functional.mse_loss(m2(a2), b2).backward()
        ref.append(m2.weight.grad)
    expected = torch.stack(ref).mean(0)
    q.put((rank, bool(torch.allclose(grad, expected, atol=1e-6))))
    dist.destroy_process_group()


def test_dataparallel_grad_accumulation_no_sync():
    """no_sync() accumulation: only the final backward is reduced, over the
    fully accumulated gradients."""
    world_size = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_accum_worker, args=(r, world_size, 29551, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world_size)]
    for p in procs:
        p.join(timeout=120)
    assert all(ok for _, ok in results), results


def _syncbn_worker(rank, world_size, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from noisynet_amd import distributed as dist_mod
    from noisynet_amd import ops

    dist_mod.init_distributed(backend="gloo")
    torch.manual_seed(7)
    full_x = torch.randn(8, 6, 5, 5)
    gamma = torch.randn(6, requires_grad=True)
    beta = torch.randn(6, requires_grad=True)
    rm = torch.zeros(6)
    rv = torch.ones(6)

    # sync path on this rank's shard
    shard = full_x[rank * 4:(rank + 1) * 4].clone().requires_grad_(True)
    y = ops.bn_act(shard, gamma, beta, rm.clone(), rv.clone(), True, 0.1,
                   1e-5, relu=True, act_max=2.0, sync=True)
    g_out = torch.ones_like(y)
    y.backward(g_out)

    # reference: single-process BN over the FULL batch
    gamma2 = gamma.detach().clone().requires_grad_(True)
    beta2 = beta.detach().clone().requires_grad_(True)
    full = full_x.clone().requires_grad_(True)
    y2 = ops.bn_act(full, gamma2, beta2, rm.clone(), rv.clone(), True, 0.1,
                    1e-5, relu=True, act_max=2.0, sync=False)
    y2.backward(torch.ones_like(y2))

    ok_fwd = torch.allclose(y, y2[rank * 4:(rank + 1) * 4], atol=1e-5)
    ok_gx = torch.allclose(shard.grad, full.grad[rank * 4:(rank + 1) * 4],
                           atol=1e-5)
    # gamma grad: local sums average to global/ws; sum across ranks == full
    gsum = gamma.grad.clone()
    dist.all_reduce(gsum)
    ok_gamma = torch.allclose(gsum, gamma2.grad, atol=1e-4)
    q.put((rank, bool(ok_fwd and ok_gx and ok_gamma),
           (ok_fwd, ok_gx, ok_gamma)))
    dist.destroy_process_group()


def test_syncbn_matches_full_batch_gloo():
    """Fused SyncBN == single-process BN over the concatenated batch."""
    world_size = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_syncbn_worker, args=(r, world_size, 29531, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world_size)]
    for p in procs:
        p.join(timeout=120)
    assert all(ok for _, ok, _ in results), results
