"""Data-parallel engine tests on CPU (gloo, world_size=2 — SURVEY §4.6)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from dwt_amd.models import LeNet


def _run_grad_parity(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dwt_amd.parallel import BucketedDataParallel
        torch.manual_seed(0)  # same init on both ranks
        model = LeNet(group_size=4)
        ddp = BucketedDataParallel(model, bucket_cap_mb=0.05)  # force many buckets
        assert ddp.enabled and len(ddp.buckets) >= 3

        # per-rank distinct data
        torch.manual_seed(100 + rank)
        x = torch.randn(8, 1, 28, 28)
        out = model(x)
        loss = out.float().pow(2).mean()
        loss.backward()

        # capture local (pre-sync is overwritten by sync, so recompute on a clone)
        torch.manual_seed(0)
        clone = LeNet(group_size=4)
        clone.load_state_dict({k: v.clone() for k, v in model.state_dict().items()})
        out_c = clone(x)
        loss_c = out_c.float().pow(2).mean()
        loss_c.backward()

        ddp.sync()

        # synced grad must equal the cross-rank average of local grads
        for (n, p), (_, pc) in zip(model.named_parameters(), clone.named_parameters()):
            local = pc.grad.clone()
            avg = local.clone()
            dist.all_reduce(avg)
            avg /= world
            assert torch.allclose(p.grad, avg, atol=1e-6), n

        if rank == 0:
            results.put("ok")
    finally:
        dist.destroy_process_group()


def _run_broadcast_and_stats(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dwt_amd.parallel import BucketedDataParallel
        torch.manual_seed(rank * 7 + 1)  # different init per rank
        model = LeNet(group_size=4)
        ddp = BucketedDataParallel(model)
        # init broadcast must have aligned parameters to rank 0's
        flat = torch.cat([p.reshape(-1) for p in model.parameters()])
        flats = [torch.zeros_like(flat) for _ in range(world)]
        dist.all_gather(flats, flat)
        assert torch.equal(flats[0], flats[1])

        # per-rank stats diverge, sync_stats averages them
        model.train()
        x = torch.randn(8, 1, 28, 28) * (1.0 + rank)
        _ = model(x)
        rm = model.wt1.running_mean.clone()
        rms = [torch.zeros_like(rm) for _ in range(world)]
        dist.all_gather(rms, rm)
        assert not torch.allclose(rms[0], rms[1])
        ddp.sync_stats()
        expect = (rms[0] + rms[1]) / 2
        assert torch.allclose(model.wt1.running_mean, expect, atol=1e-6)
        if rank == 0:
            results.put("ok")
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("fn", [_run_grad_parity, _run_broadcast_and_stats])
def test_ddp_cpu_world2(fn):
    port = 29600 + abs(hash(fn.__name__)) % 500
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=fn, args=(r, 2, port, results)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    assert results.get(timeout=5) == "ok"


def test_bucket_assembly():
    """Bucketing covers every parameter exactly once, in reverse order."""
    from dwt_amd.parallel.ddp import BucketedDataParallel
    model = LeNet(group_size=4)
    ddp = BucketedDataParallel.__new__(BucketedDataParallel)
    ddp.model = model
    ddp.params = [p for p in model.parameters() if p.requires_grad]
    ddp.buckets = []
    ddp._param_bucket = {}
    ddp.reduce_dtype = None
    ddp._build_buckets(0.1)
    seen = set()
    for b in ddp.buckets:
        for p in b.params:
            assert id(p) not in seen
            seen.add(id(p))
    assert len(seen) == len(ddp.params)
    assert ddp.buckets[0].params[0] is ddp.params[-1]  # reverse order


def test_digits_entrypoint_torchrun_cpu(tmp_path):
    """Full digits entrypoint under torchrun ws=2 (gloo) — the DP path the
    driver exercises on 8 GPUs, on CPU."""
    import subprocess, sys, os as _os
    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29712", "usps_mnist.py", "--synthetic",
         "--synthetic_size", "64", "--epochs", "1", "--group_size", "4",
         "--num_workers", "0", "--test_batch_size", "16",
         "--log_interval", "1"],
        cwd=repo, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, (r.stdout[-800:], r.stderr[-1500:])
    assert "Test set" in r.stdout


def _run_grad_parity_dtype(rank, world, port, results, dtype_name, reduce_fp32):
    """Grad parity on a small MLP at configurable world size / dtype /
    reduce mode, including the steady-state grad-as-bucket-view step
    (second backward accumulates directly into the bucket buffer)."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dwt_amd.parallel import BucketedDataParallel
        dtype = dict(float32=torch.float32, bfloat16=torch.bfloat16)[dtype_name]
        torch.manual_seed(0)
        model = torch.nn.Sequential(
            torch.nn.Linear(32, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 8)).to(dtype)
        clone = torch.nn.Sequential(
            torch.nn.Linear(32, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 8)).to(dtype)
        clone.load_state_dict(model.state_dict())
        ddp = BucketedDataParallel(
            model, bucket_cap_mb=0.01,
            reduce_dtype=torch.float32 if reduce_fp32 else None)

        for it in range(2):  # step 2 exercises the aliased-grad fast path
            torch.manual_seed(1000 * it + rank)
            x = torch.randn(8, 32).to(dtype)
            model(x).float().pow(2).mean().backward()
            clone(x).float().pow(2).mean().backward()
            ddp.sync()
            for (n, p), pc in zip(model.named_parameters(), clone.parameters()):
                avg = pc.grad.detach().float().clone()
                dist.all_reduce(avg)
                avg /= world
                tol = 1e-6 if dtype == torch.float32 else 3e-2
                assert torch.allclose(p.grad.float(), avg, atol=tol,
                                      rtol=tol), (it, n)
            if not reduce_fp32:
                # steady state: grads must alias the bucket buffers
                for b in ddp.buckets:
                    for p in b.params:
                        assert p.grad is b.views[p]
            # zero in place (FusedSGD-style) so the aliasing survives into
            # the next iteration on the fast path
            for p in model.parameters():
                if reduce_fp32:
                    p.grad = None
                elif p.grad is not None:
                    p.grad.detach().zero_()
            for p in clone.parameters():
                p.grad = None
        if rank == 0:
            results.put("ok")
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world,dtype_name,reduce_fp32", [
    (2, "float32", False),
    (4, "float32", False),
    (4, "bfloat16", False),   # pre-divided bf16 SUM
    (2, "bfloat16", True),    # fp32 bucket mode
])
def test_ddp_grad_parity_modes(world, dtype_name, reduce_fp32):
    port = 29750 + abs(hash((world, dtype_name, reduce_fp32))) % 200
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_run_grad_parity_dtype,
                         args=(r, world, port, results, dtype_name, reduce_fp32))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    assert results.get(timeout=5) == "ok"


def _run_grad_accumulation(rank, world, port, results):
    """Two accumulated backwards + one sync must equal the average of the
    SUMMED local grads (accumulate() defers the hook launches so the async
    reduce cannot race the second backward)."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dwt_amd.parallel import BucketedDataParallel
        torch.manual_seed(0)
        model = torch.nn.Linear(8, 4)
        clone = torch.nn.Linear(8, 4)
        clone.load_state_dict(model.state_dict())
        ddp = BucketedDataParallel(model, bucket_cap_mb=0.001)
        torch.manual_seed(10 + rank)
        xs = [torch.randn(4, 8) for _ in range(2)]
        with ddp.accumulate():
            model(xs[0]).pow(2).mean().backward()
        model(xs[1]).pow(2).mean().backward()
        ddp.sync()
        for x in xs:
            clone(x).pow(2).mean().backward()
        for p, pc in zip(model.parameters(), clone.parameters()):
            avg = pc.grad.clone()
            dist.all_reduce(avg)
            avg /= world
            assert torch.allclose(p.grad, avg, atol=1e-6)
        if rank == 0:
            results.put("ok")
    finally:
        dist.destroy_process_group()


def test_ddp_grad_accumulation():
    port = 29971
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_run_grad_accumulation, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    assert results.get(timeout=5) == "ok"


def _run_unused_param(rank, world, port, results):
    """A parameter whose grad never materializes (unused head) must not hang
    the bucket protocol: sync() launches the incomplete bucket with a zeroed
    segment, and the reduced grads still average correctly."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dwt_amd.parallel import BucketedDataParallel

        class TwoHead(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.body = torch.nn.Linear(16, 16)
                self.used = torch.nn.Linear(16, 4)
                self.unused = torch.nn.Linear(16, 4)

            def forward(self, x):
                return self.used(torch.relu(self.body(x)))

        torch.manual_seed(0)
        model = TwoHead()
        ddp = BucketedDataParallel(model, bucket_cap_mb=1.0)  # one bucket
        torch.manual_seed(50 + rank)
        x = torch.randn(4, 16)
        model(x).pow(2).mean().backward()
        ddp.sync()  # must not deadlock on the never-fired hooks
        g = model.body.weight.grad.clone()
        avg = g.clone()
        dist.all_reduce(avg)
        # sync() already averaged: re-averaging the averaged grad across
        # ranks must be a fixed point
        assert torch.allclose(g, avg / world, atol=1e-6)
        assert model.unused.weight.grad is not None  # view installed, zeros
        assert model.unused.weight.grad.abs().max() == 0
        if rank == 0:
            results.put("ok")
    finally:
        dist.destroy_process_group()


def test_ddp_unused_param_bucket():
    port = 29951
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_run_unused_param, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    assert results.get(timeout=5) == "ok"


def _run_stats_sync_parity(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dwt_amd.ops.functional import WhitenMulti, BatchNormMulti
        torch.manual_seed(0)
        full = torch.randn(8, 8, 5, 5, dtype=torch.float64)
        local = full[rank * 4:(rank + 1) * 4].clone().requires_grad_(True)
        ref = full.clone().requires_grad_(True)

        # whitening: synced 2-rank stats == single-process full-batch stats
        cfg = dict(parts=1, num_groups=2, eps=1e-3, momentum=0.1,
                   training=True, mode="chol", relu=False, stats_sync=True)
        out = WhitenMulti.apply(local, None, None, None, None, cfg)
        cfg1 = dict(cfg, stats_sync=False)
        out_ref = WhitenMulti.apply(ref, None, None, None, None, cfg1)
        assert torch.allclose(out, out_ref[rank * 4:(rank + 1) * 4], atol=1e-9)
        out.sum().backward()
        out_ref.sum().backward()
        assert torch.allclose(local.grad, ref.grad[rank * 4:(rank + 1) * 4],
                              atol=1e-9), (local.grad - ref.grad[rank*4:(rank+1)*4]).abs().max()

        # BN: same property
        local2 = full[rank * 4:(rank + 1) * 4].clone().requires_grad_(True)
        ref2 = full.clone().requires_grad_(True)
        bcfg = dict(parts=1, eps=1e-5, momentum=0.1, training=True,
                    relu=True, stats_sync=True)
        o2 = BatchNormMulti.apply(local2, None, None, None, None, bcfg)
        o2r = BatchNormMulti.apply(ref2, None, None, None, None,
                                   dict(bcfg, stats_sync=False))
        assert torch.allclose(o2, o2r[rank * 4:(rank + 1) * 4], atol=1e-9)
        o2.sum().backward()
        o2r.sum().backward()
        assert torch.allclose(local2.grad, ref2.grad[rank * 4:(rank + 1) * 4],
                              atol=1e-9)
        if rank == 0:
            results.put("ok")
    finally:
        dist.destroy_process_group()


def test_stats_sync_matches_full_batch():
    """stats_sync=True on 2 ranks == single-process stats over the
    concatenated batch (forward AND backward) — the 'sync' stats mode of
    SURVEY §2.3."""
    port = 29871
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_run_stats_sync_parity, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    assert results.get(timeout=5) == "ok"
