"""Intra-party process groups over RCCL (GPU) / gloo (CPU).

A *party* owns a slice of the node's MI355X GPUs, one worker process per GPU
(``torch.distributed`` backend "nccl" IS RCCL on ROCm; transport is xGMI —
7 point-to-point links × ≈153 GB/s per GPU, so ring collectives are
per-link bound and bucket sizes are chosen for that, not for NVSwitch).

Two bootstrap modes:
- **external**: the processes already exist (e.g. torchrun ranks); each calls
  :func:`init_party_group` with its rank.
- **spawned**: the party driver spawns workers via
  ``rayfed_amd.runtime.worker.DeviceWorkerPool`` which calls this internally.
"""
from __future__ import annotations

import datetime
import logging
from typing import Optional

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)


def init_party_group(
    rank: int,
    world_size: int,
    master_addr: str = "127.0.0.1",
    master_port: int = 29500,
    backend: Optional[str] = None,
    device: Optional[int] = None,
    timeout_s: float = 300.0,
) -> "dist.ProcessGroup":
    """Initialize this process's membership in its party's collective group.

    backend default: "nccl" (RCCL) when a GPU is visible, else "gloo".
    Returns the default process group.
    """
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if device is not None and torch.cuda.is_available():
        torch.cuda.set_device(device)
    store = dist.TCPStore(
        master_addr,
        master_port,
        world_size,
        is_master=(rank == 0),
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    dist.init_process_group(
        backend=backend,
        store=store,
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    logger.info(
        "party group up: rank %d/%d backend=%s", rank, world_size, backend
    )
    return dist.group.WORLD


def destroy_party_group() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
