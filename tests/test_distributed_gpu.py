"""RCCL-on-hardware tests (single MI355X box).

Exercises the actual nccl(=RCCL) backend that the 8-GPU scaling bench will
use -- init/teardown at world_size=1, and a 2-ranks-on-one-GPU rehearsal of
the bucketed overlap all-reduce, bf16 buckets and SyncBN (reference
capability: main.py:737-803). The multi-rank test shares the single device
between both processes; if this RCCL build refuses co-located ranks the
test reports that explicitly instead of passing vacuously.
"""

import multiprocessing as mp
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_rccl_init_world_size_1():
    """init_process_group('nccl') + all_reduce + barrier on one rank."""
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29561"
    dist.init_process_group(backend="nccl", init_method="env://",
                            rank=0, world_size=1)
    try:
        # values exactly representable in bf16 (ws=1 all_reduce == identity)
        ref = torch.arange(1024, dtype=torch.float32).to(torch.bfloat16)
        t = ref.clone().to("cuda")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert torch.equal(t.cpu(), ref)
        dist.barrier()
    finally:
        dist.destroy_process_group()


def _two_rank_worker(rank, world_size, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = "0"  # both ranks share the single GPU
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        import torch.distributed as dist

        from noisynet_amd import distributed as dist_mod
        from noisynet_amd import ops

        dist_mod.init_distributed(backend="nccl", timeout_s=90)
        device = torch.device("cuda", 0)
        torch.manual_seed(1234)
        model = torch.nn.Sequential(
            torch.nn.Linear(64, 128), torch.nn.ReLU(),
            torch.nn.Linear(128, 10)).to(device).bfloat16()
        dp = dist_mod.DataParallel(model, bucket_cap_mb=1)

        torch.manual_seed(500 + rank)
        x = torch.randn(16, 64, device=device, dtype=torch.bfloat16)
        y = torch.randint(0, 10, (16,), device=device)
        loss = ops.cross_entropy(model(x), y)
        loss.backward()
        dp.finish()
        grad = model[0].weight.grad.float().cpu().clone()

        # serial reference: average of both ranks' grads
        ref = []
        for r in range(world_size):
            torch.manual_seed(1234)
            m2 = torch.nn.Sequential(
                torch.nn.Linear(64, 128), torch.nn.ReLU(),
                torch.nn.Linear(128, 10)).to(device).bfloat16()
            torch.manual_seed(500 + r)
            xr = torch.randn(16, 64, device=device, dtype=torch.bfloat16)
            yr = torch.randint(0, 10, (16,), device=device)
            ops.cross_entropy(m2(xr), yr).backward()
            ref.append(m2[0].weight.grad.float().cpu())
        expected = torch.stack(ref).mean(0)
        ok = torch.allclose(grad, expected, atol=2e-3)

        # SyncBN forward over RCCL
        bn_x = torch.randn(8, 6, 5, 5, device=device)
        gamma = torch.ones(6, device=device)
        beta = torch.zeros(6, device=device)
        rm, rv = torch.zeros(6, device=device), torch.ones(6, device=device)
        yb = ops.bn_act(bn_x, gamma, beta, rm, rv, True, 0.1, 1e-5,
                        relu=True, act_max=0.0, sync=True)
        torch.cuda.synchronize()
        ok_bn = bool(torch.isfinite(yb).all())

        q.put((rank, "ok" if (ok and ok_bn) else
               "mismatch grad=%s bn=%s" % (ok, ok_bn)))
        dist.destroy_process_group()
    except Exception as exc:  # surface the failure mode to the parent
        q.put((rank, "error: %r" % (exc,)))


def test_rccl_two_ranks_one_gpu():
    """Bucketed bf16 all-reduce + SyncBN with 2 RCCL ranks on one device."""
    world_size = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_two_rank_worker,
                         args=(r, world_size, 29562, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world_size)]
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    msgs = {r: m for r, m in results}
    dup = [m for m in msgs.values()
           if "error" in m and ("uplicate" in m or "same device" in m)]
    if dup:
        pytest.skip("RCCL build refuses co-located ranks: %s" % dup[0])
    assert all(m == "ok" for m in msgs.values()), msgs
