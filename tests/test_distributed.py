"""Multi-process distributed-path tests (gloo backend, world_size=2, CPU).

The reference never simulates multi-GPU; here the DDP loop,
gradient sync, scalar reductions, and the sharded sampler are covered on
CPU so the RCCL path is correct by construction before it reaches a GPU.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _init(rank, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)


def _run_reduce(rank, port, q):
    try:
        _init(rank, port)
        from esr_amd.parallel import reduce_tensor, reduce_dict
        t = torch.tensor([float(rank + 1)])
        out = reduce_tensor(t)                      # mean(1, 2) = 1.5
        d = reduce_dict({"a": torch.tensor(float(rank)),
                         "b": torch.tensor(2.0 * rank)})
        q.put((rank, out.item(), d["a"].item(), d["b"].item()))
    finally:
        dist.destroy_process_group()


def _run_ddp_grads(rank, port, q):
    try:
        _init(rank, port)
        from esr_amd.models import build_model
        from esr_amd.parallel import wrap_ddp
        torch.manual_seed(0)  # same init on both ranks
        model = build_model("ESRNet", inch=2, basech=4, num_frame=3)
        model = wrap_ddp(model)
        torch.manual_seed(100 + rank)  # different data per rank
        x = torch.randn(1, 3, 2, 16, 16)
        model.module.reset_states()
        loss = (model(x) ** 2).mean()
        loss.backward()
        g = next(p.grad for p in model.parameters() if p.grad is not None)
        q.put((rank, g.flatten()[:5].tolist()))
    finally:
        dist.destroy_process_group()


def _run_sampler(rank, port, q):
    try:
        _init(rank, port)
        from torch.utils.data.distributed import DistributedSampler
        ds = list(range(10))
        s = DistributedSampler(ds, shuffle=False)
        q.put((rank, list(iter(s))))
    finally:
        dist.destroy_process_group()


def _spawn(fn, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=fn, args=(r, port, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, *vals = q.get(timeout=300)
        results[rank] = vals
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    return results


@pytest.mark.timeout(400)
def test_reduce_tensor_and_dict():
    res = _spawn(_run_reduce, 29511)
    for rank in range(WORLD):
        mean, a, b = res[rank]
        assert abs(mean - 1.5) < 1e-6
        assert abs(a - 0.5) < 1e-6          # mean(0, 1)
        assert abs(b - 1.0) < 1e-6          # mean(0, 2)


@pytest.mark.timeout(400)
def test_ddp_gradient_sync():
    res = _spawn(_run_ddp_grads, 29513)
    g0, g1 = res[0][0], res[1][0]
    assert g0 == pytest.approx(g1, abs=1e-6), "DDP grads differ across ranks"


@pytest.mark.timeout(400)
def test_distributed_sampler_partitions():
    res = _spawn(_run_sampler, 29515)
    idx0, idx1 = set(res[0][0]), set(res[1][0])
    assert len(idx0) == len(idx1) == 5
    assert idx0.isdisjoint(idx1)


def _run_watchdog(rank, port, q):
    try:
        _init(rank, port)
        from esr_amd.parallel.watchdog import Watchdog
        events = []
        wd = Watchdog(timeout=2.0, interval=0.2,
                      on_failure=lambda ranks: events.append(ranks))
        wd.beat()
        import time
        if rank == 0:
            # rank 0 keeps beating; rank 1 goes silent after one beat
            wd.start()
            time.sleep(4.0)
            wd.stop()
            q.put((rank, list(wd.failed_ranks), len(events)))
        else:
            time.sleep(4.5)   # silent: no further beats
            q.put((rank, [], 0))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(400)
def test_watchdog_detects_silent_rank():
    res = _spawn(_run_watchdog, 29517)
    failed, n_events = res[0]
    assert failed == [1], f"rank 0 should flag rank 1, got {failed}"
    assert n_events >= 1


def _run_eager_fallback_sync(rank, port, q):
    """Simulates the hip_graphs runtime-fallback: model NOT DDP-wrapped,
    eager backward + Trainer._sync_grads_if_needed must still converge the
    ranks' gradients (advisor finding r1)."""
    try:
        _init(rank, port)
        from esr_amd.models import build_model
        torch.manual_seed(0)
        model = build_model("ESRNet", inch=2, basech=4, num_frame=3)
        torch.manual_seed(200 + rank)
        x = torch.randn(1, 3, 2, 16, 16)
        model.reset_states()
        loss = (model(x) ** 2).mean()
        loss.backward()

        class _T:  # minimal Trainer stand-in exposing the mixin pieces
            pass
        from esr_amd.engine.trainer import Trainer
        t = _T()
        t.model = model
        Trainer._sync_grads_if_needed(t)
        g = next(p.grad for p in model.parameters() if p.grad is not None)
        q.put((rank, g.flatten()[:5].tolist()))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(400)
def test_eager_fallback_grad_sync():
    res = _spawn(_run_eager_fallback_sync, 29519)
    assert res[0][0] == pytest.approx(res[1][0], abs=1e-6), \
        "eager fallback left per-rank gradients unsynchronized"
