"""Intra-party parallelism on CPU: gloo world_size=2 collectives + the
device-worker pool.  The same code paths run RCCL on MI355X (backend picks
nccl when a GPU is visible)."""
import multiprocessing

import pytest

torch = pytest.importorskip("torch")

from tests._util import free_ports  # noqa: E402

_mp = multiprocessing.get_context("spawn")


def _allreduce_worker(rank, world, port, q):
    import torch

    from rayfed_amd.parallel.fedavg import BucketedAllReducer
    from rayfed_amd.parallel.group import destroy_party_group, init_party_group

    init_party_group(rank, world, master_port=port, backend="gloo")
    torch.manual_seed(rank)
    tensors = [torch.randn(1000), torch.randn(64, 32), torch.randn(7)]
    # Deterministic reference: rank r's tensor j = seeded randn
    refs = []
    for j, shape in enumerate([(1000,), (64, 32), (7,)]):
        acc = torch.zeros(shape)
        for r in range(world):
            torch.manual_seed(r)
            ts = [torch.randn(1000), torch.randn(64, 32), torch.randn(7)]
            acc += ts[j]
        refs.append(acc / world)

    reducer = BucketedAllReducer(bucket_bytes=8192, average=True)
    reducer.allreduce_(tensors)
    ok = all(torch.allclose(t, ref, atol=1e-5) for t, ref in zip(tensors, refs))
    q.put((rank, ok))
    destroy_party_group()


def test_bucketed_allreduce_gloo_world2():
    port = free_ports(1)[0]
    q = _mp.Queue()
    procs = [
        _mp.Process(target=_allreduce_worker, args=(r, 2, port, q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, ok = q.get(timeout=120)
        results[rank] = ok
    for p in procs:
        p.join(timeout=30)
    assert results == {0: True, 1: True}


def test_weighted_combine_cpu_reference():
    from rayfed_amd.parallel.fedavg import weighted_combine_

    torch.manual_seed(0)
    ins = [torch.randn(1000) for _ in range(3)]
    out = torch.empty(1000)
    weighted_combine_(out, ins, [0.5, 0.25, 0.25])
    ref = 0.5 * ins[0] + 0.25 * ins[1] + 0.25 * ins[2]
    assert torch.allclose(out, ref, atol=1e-6)


# -- device worker pool (CPU devices) -----------------------------------------
def _square(x):
    return x * x


def _make_tensor(n):
    import torch

    return torch.arange(n, dtype=torch.float32)


def _sum_handle(t):
    return float(t.sum())


def test_worker_pool_tasks_and_handles():
    from rayfed_amd.runtime.worker import DeviceWorkerPool

    pool = DeviceWorkerPool(devices=[None, None])
    try:
        assert pool.submit(0, _square, (7,)).result(60) == 49
        assert pool.submit(1, _square, (8,)).result(60) == 64
        # keep=True: result stays in the worker; fetch pulls it back.
        h = pool.submit(0, _make_tensor, (10,), keep=True).result(60)
        assert h.worker_id == 0
        s = pool.submit(0, _sum_handle, (h,)).result(60)
        assert s == float(sum(range(10)))
        fetched = pool.fetch(h).result(60)
        assert fetched.shape == (10,)
        pool.delete(h).result(60)
    finally:
        pool.shutdown()


def _fail():
    raise ValueError("worker boom")


def test_worker_pool_error_propagates():
    from rayfed_amd.runtime.worker import DeviceWorkerPool

    pool = DeviceWorkerPool(devices=[None])
    try:
        ref = pool.submit(0, _fail)
        with pytest.raises(ValueError, match="worker boom"):
            ref.result(60)
    finally:
        pool.shutdown()


def _rank_allreduce(x):
    import torch
    import torch.distributed as dist

    t = torch.tensor([float(x)])
    dist.all_reduce(t)
    return float(t.item())


def test_worker_pool_with_party_group_collective():
    """submit_all runs the same collective on every worker (gloo, CPU) —
    the pattern FedAvg uses over RCCL on the GPU node."""
    from rayfed_amd.runtime.worker import DeviceWorkerPool

    pool = DeviceWorkerPool(
        devices=[None, None], with_party_group=True, backend="gloo"
    )
    try:
        refs = pool.submit_all(_rank_allreduce, (5,))
        vals = [r.result(120) for r in refs]
        assert vals == [10.0, 10.0]
    finally:
        pool.shutdown()


def _flat_allreduce_worker(rank, world, port, q):
    import torch

    from rayfed_amd.parallel.fedavg import allreduce_flat_
    from rayfed_amd.parallel.group import destroy_party_group, init_party_group

    init_party_group(rank, world, master_port=port, backend="gloo")
    torch.manual_seed(rank)
    flat = torch.randn(100_000)
    ref_inputs = []
    for r in range(world):
        torch.manual_seed(r)
        ref_inputs.append(torch.randn(100_000))
    expect = sum(ref_inputs) / world
    allreduce_flat_(flat, bucket_bytes=64 * 1024)
    q.put((rank, bool(torch.allclose(flat, expect, atol=1e-5))))
    destroy_party_group()


def test_allreduce_flat_gloo_world2():
    """In-place chunked all-reduce of a flat grad buffer (bench fedavg path)."""
    port = free_ports(1)[0]
    q = _mp.Queue()
    procs = [
        _mp.Process(target=_flat_allreduce_worker, args=(r, 2, port, q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, ok = q.get(timeout=120)
        results[rank] = ok
    for p in procs:
        p.join(timeout=30)
    assert results == {0: True, 1: True}


def _die():
    import os

    os._exit(17)


def test_worker_death_fails_outstanding_futures():
    """A crashed worker must fail its pending refs, not hang callers."""
    from rayfed_amd.runtime.worker import DeviceWorkerPool

    pool = DeviceWorkerPool(devices=[None])
    try:
        ref = pool.submit(0, _die)
        with pytest.raises(RuntimeError, match="exited unexpectedly"):
            ref.result(60)
    finally:
        pool.shutdown()
