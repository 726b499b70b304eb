#!/usr/bin/env python3
"""Flagship benchmark — BASELINE.json headline metric on MI355X.

Metric: cross-party objects/sec of the many_tiny_tasks harness
(/root/reference/benchmarks/many_tiny_tasks_benchmark.py:44-59 — per
iteration: one actor task per party + one cross-party aggregate + one
fed.get broadcast → 2 cross-party object transfers), measured on this
engine's control path.

Scaling model (``--gpus N``, weak): N GPUs are split across the two parties
(alice = GPUs [0, N/2), bob = the rest).  Each GPU pair (i, i+N/2) runs one
federation *lane* — an independent 2-party fed job on its own loopback ports
— so per-GPU work is fixed while whole-job objects/sec grows with N.
N=1 runs both parties of one lane on GPU 0 (bob in a forked subprocess).

Launch contract (driver): ``python bench.py --gpus N --steps K --warmup W``;
for N>1 via ``python -m torch.distributed.run --nproc-per-node N`` — one
rank per GPU, rank r < N/2 drives alice of lane r, rank r >= N/2 drives bob
of lane r-N/2.  Rank 0 prints ONE JSON line; elapsed is the MAX over ranks.

Modes: ``tiny`` (headline), ``push`` (BASELINE config 3: 4 GiB bf16 tensor
push alice→bob, GB/s — reported in the JSON config block when run).
"""
from __future__ import annotations

import argparse
import json
import multiprocessing
import os
import sys
import time


def _has_cuda():
    try:
        import torch

        return torch.cuda.is_available()
    except ImportError:
        return False


# ------------------------------------------------------------------ tiny mode
def _tiny_driver(party: str, addresses, steps: int, warmup: int, device: int,
                 job_name: str, barrier_cb=None, result_q=None):
    """The many_tiny_tasks loop; identical code runs in both parties."""
    import rayfed_amd as fed

    use_gpu = _has_cuda()
    if use_gpu:
        import torch

        torch.cuda.set_device(device)

    fed.init(addresses=addresses, party=party, job_name=job_name,
             logging_level="warning")

    @fed.remote
    class MyActor:
        def __init__(self, device, use_gpu):
            self._use_gpu = use_gpu
            if use_gpu:
                import torch

                self._t = torch.zeros(256, device=f"cuda:{device}")

        def run(self):
            if self._use_gpu:
                self._t += 1.0  # a real (tiny) HIP kernel per task
            return 1

    @fed.remote
    class Aggregator:
        def aggr(self, v1, v2):
            return v1 + v2

    actor_alice = MyActor.party("alice").remote(device, use_gpu)
    actor_bob = MyActor.party("bob").remote(device, use_gpu)
    aggregator = Aggregator.party("alice").remote()

    def one_iter():
        va = actor_alice.run.remote()
        vb = actor_bob.run.remote()
        s = aggregator.aggr.remote(va, vb)
        return fed.get(s)

    for _ in range(warmup):
        assert one_iter() == 2
    if use_gpu:
        import torch

        torch.cuda.synchronize()
    if barrier_cb is not None:
        barrier_cb()
    t0 = time.perf_counter()
    for _ in range(steps):
        one_iter()
    if use_gpu:
        import torch

        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if barrier_cb is not None:
        barrier_cb()
    fed.shutdown()
    if result_q is not None:
        result_q.put(elapsed)
    return elapsed


# ------------------------------------------------------------------ push mode
def _push_driver(party: str, addresses, steps: int, warmup: int, device: int,
                 job_name: str, nbytes: int, barrier_cb=None, result_q=None):
    """BASELINE config 3: bf16 tensor push alice→bob; end-to-end GB/s."""
    import torch

    import rayfed_amd as fed

    use_gpu = _has_cuda()
    dev = f"cuda:{device}" if use_gpu else "cpu"
    if use_gpu:
        torch.cuda.set_device(device)
    fed.init(addresses=addresses, party=party, job_name=job_name,
             logging_level="warning")

    numel = nbytes // 2

    @fed.remote
    class Producer:
        def __init__(self, dev, numel):
            self._t = torch.randn(numel, dtype=torch.bfloat16, device=dev)

        def produce(self):
            return self._t

    @fed.remote
    class Consumer:
        def consume(self, t):
            return int(t.numel() * t.element_size())

    producer = Producer.party("alice").remote(dev, numel)
    consumer = Consumer.party("bob").remote()

    def one_iter():
        t = producer.produce.remote()
        n = consumer.consume.remote(t)
        return fed.get(n)

    for _ in range(warmup):
        assert one_iter() == nbytes
    if use_gpu:
        torch.cuda.synchronize()
    if barrier_cb is not None:
        barrier_cb()
    t0 = time.perf_counter()
    for _ in range(steps):
        one_iter()
    if use_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if barrier_cb is not None:
        barrier_cb()
    fed.shutdown()
    if result_q is not None:
        result_q.put(elapsed)
    return elapsed


_DRIVERS = {"tiny": _tiny_driver, "push": _push_driver}


def _run_single_process(mode, steps, warmup, push_bytes) -> float:
    """N=1: alice in-process, bob forked, both on GPU 0."""
    from tests._util import make_addresses  # free-port helper

    addresses = make_addresses(["alice", "bob"])
    ctx = multiprocessing.get_context("fork")
    kwargs = {}
    args_extra = (push_bytes,) if mode == "push" else ()
    bob = ctx.Process(
        target=_DRIVERS[mode],
        args=("bob", addresses, steps, warmup, 0, f"bench_{mode}") + args_extra,
    )
    bob.start()
    elapsed = _DRIVERS[mode](
        "alice", addresses, steps, warmup, 0, f"bench_{mode}", *args_extra
    )
    bob.join(timeout=600)
    if bob.is_alive():
        bob.terminate()
        raise RuntimeError("bob party hung")
    return elapsed


def _run_torchrun(mode, steps, warmup, push_bytes, rank, world, local_rank):
    import torch.distributed as dist

    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    lanes = world // 2
    lane = rank % lanes
    party = "alice" if rank < lanes else "bob"
    base = int(os.environ.get("RAYFED_BENCH_BASE_PORT", "23500"))
    addresses = {
        "alice": f"127.0.0.1:{base + lane * 2}",
        "bob": f"127.0.0.1:{base + lane * 2 + 1}",
    }

    def barrier():
        dist.barrier()

    args_extra = (push_bytes,) if mode == "push" else ()
    elapsed = _DRIVERS[mode](
        party, addresses, steps, warmup, local_rank, f"bench_{mode}_lane{lane}",
        *args_extra, barrier_cb=barrier,
    )
    # MAX over ranks (the contract).
    import torch

    t = torch.tensor([elapsed], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    dist.destroy_process_group()
    return float(t.item()), lanes


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=2000)
    p.add_argument("--warmup", type=int, default=200)
    p.add_argument("--mode", choices=["tiny", "push"], default="tiny")
    p.add_argument("--push-gib", type=float, default=4.0,
                   help="tensor size for --mode push (GiB)")
    args = p.parse_args()

    if args.mode == "push" and args.steps > 50:
        # 4 GiB per step: keep the default run under minutes.
        args.steps = min(args.steps, 20)
        args.warmup = min(args.warmup, 3)
    push_bytes = int(args.push_gib * (1 << 30))

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    if world > 1:
        assert world % 2 == 0, "world size must be even (2 parties)"
        elapsed, lanes = _run_torchrun(
            args.mode, args.steps, args.warmup, push_bytes, rank, world, local_rank
        )
        if rank != 0:
            return
        n_gpus = world
    else:
        elapsed = _run_single_process(args.mode, args.steps, args.warmup, push_bytes)
        lanes, n_gpus = 1, args.gpus

    ms_per_step = elapsed * 1000.0 / args.steps
    if args.mode == "tiny":
        # 2 cross-party transfers per iteration per lane (value + broadcast).
        value = 2.0 * args.steps * lanes / elapsed
        metric = "cross_party_objects_per_sec"
        unit = "objects/s"
        config = {
            "model": "many_tiny_tasks (2-party aggregate loop)",
            "global_batch": args.steps * lanes,
            "seq_len": 1,
            "parallelism": f"fed2p-weak x{lanes} lanes",
            "per_task_overhead_ms": ms_per_step,
        }
    else:
        value = push_bytes * args.steps * lanes / elapsed / 1e9
        metric = "cross_party_tensor_push_GBps"
        unit = "GB/s"
        config = {
            "model": f"{args.push_gib} GiB bf16 tensor push alice->bob",
            "global_batch": args.steps,
            "seq_len": push_bytes // 2,
            "parallelism": f"fed2p-weak x{lanes} lanes",
        }

    print(json.dumps({
        "metric": metric,
        "value": round(value, 3),
        "unit": unit,
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 4),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": config,
    }))


if __name__ == "__main__":
    main()
