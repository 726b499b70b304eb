"""Bucket-size sweep for the intra-party all-reduce over RCCL/xGMI.

SURVEY §7 hard parts: ring collectives are per-link bound on xGMI
(7 x ~153 GB/s point-to-point links per MI355X), so the right bucket size
is a measurement, not an assumption.  Run on an 8-GPU node:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 --master-port 29520 scripts/xgmi_sweep.py

Prints per-bucket-size bus bandwidth for a Llama-3-8B-sized flat gradient
(rank 0).  Works on CPU/gloo too (world>=2) for plumbing checks.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.distributed as dist

from rayfed_amd.parallel.fedavg import allreduce_flat_


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend="nccl" if use_gpu else "gloo",
                            rank=rank, world_size=world)
    n = (1 << 33) if use_gpu else (1 << 24)  # 16 GiB bf16 / 32 MiB cpu
    flat = torch.empty(n, dtype=torch.bfloat16,
                       device=f"cuda:{local_rank}" if use_gpu else "cpu")
    flat.uniform_(-1, 1)
    nbytes = n * 2
    for bucket_mb in (32, 64, 128, 256, 512):
        for _ in range(2):  # warm
            allreduce_flat_(flat, bucket_bytes=bucket_mb << 20)
        if use_gpu:
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        iters = 5
        for _ in range(iters):
            allreduce_flat_(flat, bucket_bytes=bucket_mb << 20)
        if use_gpu:
            torch.cuda.synchronize()
        dist.barrier()
        dt = (time.perf_counter() - t0) / iters
        # Ring all-reduce bus bandwidth convention: 2*(W-1)/W * bytes / t.
        bus = 2 * (world - 1) / world * nbytes / dt / 1e9
        if rank == 0:
            print(f"bucket {bucket_mb:4d} MiB: {dt*1e3:8.1f} ms/iter, "
                  f"bus bandwidth {bus:7.1f} GB/s", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
