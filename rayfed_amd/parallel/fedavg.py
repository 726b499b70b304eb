"""FedAvg aggregation: bucketed intra-party all-reduce + HIP weighted reduce.

The FedAvg round on MI355X (BASELINE config 4 — Llama-3-8B gradients,
4 GPUs/party):

  1. every party worker holds its local gradient shard (bf16);
  2. **intra-party**: bucketed all-reduce over RCCL/xGMI averages gradients
     across the party's GPUs (this module);
  3. **cross-party**: the party leader packs the averaged gradients through
     the GPU data plane and pushes them over the cross-silo transport; the
     receiving party's weighted combine uses the HIP ``fedavg_reduce_``
     kernel (csrc/rayfed_hip.hip).

Bucketing is sized for xGMI: ring all-reduce is per-link bound
(≈153 GB/s/link), so we want few, large buckets — default 128 MiB — and we
overlap bucket copies with the in-flight collective via async ops.
"""
from __future__ import annotations

import logging
from typing import Iterable, List, Optional, Sequence

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

DEFAULT_BUCKET_BYTES = 128 << 20


class BucketedAllReducer:
    """All-reduce a list of tensors in pre-allocated flat buckets with
    copy/collective overlap (async ops pipelined across buckets)."""

    def __init__(
        self,
        bucket_bytes: int = DEFAULT_BUCKET_BYTES,
        group: Optional["dist.ProcessGroup"] = None,
        average: bool = True,
    ):
        self._bucket_bytes = bucket_bytes
        self._group = group
        self._average = average
        self._flat_cache: dict = {}

    def _flat(self, numel: int, dtype, device) -> torch.Tensor:
        key = (numel, dtype, str(device))
        buf = self._flat_cache.get(key)
        if buf is None:
            buf = torch.empty(numel, dtype=dtype, device=device)
            self._flat_cache[key] = buf
        return buf

    def allreduce_(self, tensors: Sequence[torch.Tensor]) -> None:
        """In-place all-reduce (sum or mean) of ``tensors`` across the group."""
        if not tensors:
            return
        world = dist.get_world_size(self._group)
        if world == 1:
            return

        # Partition into buckets of ~bucket_bytes, same dtype per bucket.
        buckets: List[List[torch.Tensor]] = []
        cur: List[torch.Tensor] = []
        cur_bytes = 0
        cur_dtype = None
        for t in tensors:
            tb = t.numel() * t.element_size()
            if cur and (cur_bytes + tb > self._bucket_bytes or t.dtype != cur_dtype):
                buckets.append(cur)
                cur, cur_bytes = [], 0
            cur.append(t)
            cur_bytes += tb
            cur_dtype = t.dtype
        if cur:
            buckets.append(cur)

        # Pipeline: pack bucket i+1 while bucket i's collective is in flight.
        pending = []  # (work, flat, bucket)
        for bucket in buckets:
            numel = sum(t.numel() for t in bucket)
            flat = self._flat(numel, bucket[0].dtype, bucket[0].device)
            off = 0
            for t in bucket:
                flat[off : off + t.numel()].copy_(t.view(-1))
                off += t.numel()
            work = dist.all_reduce(flat, op=dist.ReduceOp.SUM,
                                   group=self._group, async_op=True)
            pending.append((work, flat, bucket))
            # Keep at most 2 collectives in flight: enough to overlap the
            # next bucket's pack copy with the current ring transfer.
            if len(pending) > 2:
                self._drain_one(pending.pop(0), world)
        while pending:
            self._drain_one(pending.pop(0), world)

    def _drain_one(self, item, world: int) -> None:
        work, flat, bucket = item
        work.wait()
        if self._average:
            flat.div_(world)
        off = 0
        for t in bucket:
            t.view(-1).copy_(flat[off : off + t.numel()])
            off += t.numel()


def allreduce_flat_(
    flat: torch.Tensor,
    bucket_bytes: int = DEFAULT_BUCKET_BYTES,
    group: Optional["dist.ProcessGroup"] = None,
    average: bool = True,
    max_inflight: int = 2,
) -> None:
    """In-place bucketed all-reduce of an already-flat gradient buffer
    (the flat-grads layout real DDP stacks keep): no pack/unpack copies at
    all — chunk views of ``flat`` reduce directly, pipelined ``max_inflight``
    deep for xGMI ring overlap."""
    world = dist.get_world_size(group)
    if world == 1 or flat.numel() == 0:
        return
    step = max(1, bucket_bytes // flat.element_size())
    pending = []
    for lo in range(0, flat.numel(), step):
        chunk = flat[lo : lo + step]
        pending.append(
            (dist.all_reduce(chunk, op=dist.ReduceOp.SUM, group=group,
                             async_op=True), chunk)
        )
        if len(pending) > max_inflight:
            work, done_chunk = pending.pop(0)
            work.wait()
            if average:
                done_chunk.div_(world)
    for work, chunk in pending:
        work.wait()
        if average:
            chunk.div_(world)


def allreduce_gradients(
    params_or_grads: Iterable[torch.Tensor],
    bucket_bytes: int = DEFAULT_BUCKET_BYTES,
    group: Optional["dist.ProcessGroup"] = None,
    average: bool = True,
) -> None:
    """Convenience one-shot: average .grad fields (or raw tensors) in-place
    across the party group."""
    grads = []
    for p in params_or_grads:
        g = p.grad if hasattr(p, "grad") and p.grad is not None else p
        grads.append(g)
    BucketedAllReducer(bucket_bytes, group, average).allreduce_(grads)


def weighted_combine_(
    out: torch.Tensor,
    inputs: List[torch.Tensor],
    weights: List[float],
) -> torch.Tensor:
    """Cross-party FedAvg combine: out = Σ w_i · in_i.

    On GPU this is the HIP ``fedavg_reduce_`` kernel (fp32 accumulation,
    16-byte lanes); on CPU a float32 torch reference with identical
    numerics contract.  A :class:`~rayfed_amd.ops.gpu_plane.LazyIpcTensor`
    input (zero-copy receive, config lazy_ipc) dispatches the fused
    combine+verify kernel reading the peer's slabs directly — supported for
    the 2-input bf16 case (one lazy + one resident tensor).
    """
    from rayfed_amd.ops.gpu_plane import LazyIpcTensor

    lazy_idx = [i for i, t in enumerate(inputs) if isinstance(t, LazyIpcTensor)]
    if lazy_idx:
        if (
            len(inputs) == 2
            and len(lazy_idx) == 1
            and out.is_cuda
            and out.dtype == torch.bfloat16
        ):
            li = lazy_idx[0]
            lazy, local = inputs[li], inputs[1 - li]
            lazy.combine_into(out, local.view(-1), weights[1 - li], weights[li])
            return out
        # General case: materialize the lazies and fall through.
        inputs = [
            t.materialize() if isinstance(t, LazyIpcTensor) else t
            for t in inputs
        ]
    if out.is_cuda:
        from rayfed_amd.ops import _hip_loader

        _hip_loader.load().fedavg_reduce_(out, list(inputs), list(weights))
        return out
    acc = torch.zeros(out.shape, dtype=torch.float32)
    for w, t in zip(weights, inputs):
        acc += float(w) * t.float()
    out.copy_(acc.to(out.dtype))
    return out
