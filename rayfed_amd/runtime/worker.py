"""Device worker pool: one process per GPU inside a party.

The moral equivalent of Ray's per-node worker processes (SURVEY.md §7 step 1
"per-party controller process + worker processes (one per GPU) with an object
table"), built MI355X-first:

- each worker pins one HIP device and (optionally) a rank in the party's
  RCCL communicator over xGMI (``rayfed_amd.parallel.group``);
- results can stay **device-resident** in the owning worker
  (``keep=True`` → a :class:`RemoteHandle` into the worker's object table),
  so a FedAvg round never round-trips gradients through the driver;
- collectives are driver-orchestrated: the same callable is submitted to
  every worker, which executes it under its own rank (e.g. a bucketed
  all-reduce from ``rayfed_amd.parallel.fedavg``).

Transport driver↔worker is a multiprocessing Pipe with cloudpickle payloads
(control-plane only — tensors cross GPUs via RCCL, never via the pipe unless
explicitly fetched).
"""
from __future__ import annotations

import itertools
import logging
import multiprocessing as mp
import threading
import traceback
from concurrent.futures import Future
from typing import Any, Dict, List, Optional, Sequence

import cloudpickle

from rayfed_amd.runtime.object_ref import ObjectRef

logger = logging.getLogger(__name__)


class RemoteHandle:
    """A reference to an object living in a worker's object table."""

    __slots__ = ("worker_id", "obj_id")

    def __init__(self, worker_id: int, obj_id: int):
        self.worker_id = worker_id
        self.obj_id = obj_id

    def __repr__(self):
        return f"RemoteHandle(w{self.worker_id}, o{self.obj_id})"


def _worker_main(worker_id: int, device: Optional[int], conn,
                 group_cfg: Optional[dict],
                 kv_snapshot: Optional[dict] = None):
    """Worker process loop: execute tasks, keep an object table."""
    import torch

    if device is not None and torch.cuda.is_available():
        torch.cuda.set_device(device)
    if kv_snapshot:
        # Seed this process's internal KV with the driver's job/cluster
        # config so tasks can read fed_config.get_cluster_config() etc.
        # (the reference gets this via Ray's GCS KV — compatible_utils.py).
        from rayfed_amd._private import kv as kv_mod

        k = kv_mod._init_internal_kv(kv_snapshot["job_name"])
        for key, value in kv_snapshot["entries"].items():
            if value is not None:
                k.put(key, value)
    if group_cfg is not None:
        from rayfed_amd.parallel.group import init_party_group

        init_party_group(
            rank=worker_id,
            world_size=group_cfg["world_size"],
            master_addr=group_cfg.get("master_addr", "127.0.0.1"),
            master_port=group_cfg["master_port"],
            backend=group_cfg.get("backend"),
            device=device,
        )

    store: Dict[int, Any] = {}
    obj_counter = itertools.count(1)

    def resolve(x):
        if isinstance(x, RemoteHandle):
            return store[x.obj_id]
        if isinstance(x, (list, tuple)):
            return type(x)(resolve(v) for v in x)
        if isinstance(x, dict):
            return {k: resolve(v) for k, v in x.items()}
        return x

    while True:
        try:
            msg = conn.recv_bytes()
        except (EOFError, OSError):
            break
        op, task_id, payload = cloudpickle.loads(msg)
        if op == "stop":
            break
        try:
            if op == "task":
                fn, args, kwargs, keep = payload
                args = resolve(args)
                kwargs = resolve(kwargs)
                result = fn(*args, **kwargs)
                if keep:
                    oid = next(obj_counter)
                    store[oid] = result
                    reply = (task_id, False, RemoteHandle(worker_id, oid))
                else:
                    reply = (task_id, False, result)
            elif op == "fetch":
                reply = (task_id, False, store[payload])
            elif op == "delete":
                store.pop(payload, None)
                reply = (task_id, False, None)
            else:
                reply = (task_id, True, ValueError(f"bad op {op!r}"))
        except BaseException as e:  # noqa: BLE001
            logger.debug("worker %d task failed:\n%s", worker_id,
                         traceback.format_exc())
            reply = (task_id, True, e)
        try:
            conn.send_bytes(cloudpickle.dumps(reply))
        except Exception:  # pickling the result/exception failed
            conn.send_bytes(
                cloudpickle.dumps(
                    (task_id, True, RuntimeError("unpicklable task result"))
                )
            )
    if group_cfg is not None:
        from rayfed_amd.parallel.group import destroy_party_group

        destroy_party_group()


class DeviceWorkerPool:
    """Spawn + drive one worker process per device."""

    def __init__(
        self,
        devices: Sequence[Optional[int]],
        with_party_group: bool = False,
        master_port: Optional[int] = None,
        backend: Optional[str] = None,
        start_method: str = "spawn",
        kv_snapshot: Optional[dict] = None,
    ):
        ctx = mp.get_context(start_method)
        self._task_counter = itertools.count(1)
        self._futures: Dict[int, Future] = {}
        self._fut_lock = threading.Lock()
        self._workers: List = []
        self._conns: List = []
        self._send_locks: List[threading.Lock] = []
        group_cfg = None
        if with_party_group:
            if master_port is None:
                import socket

                s = socket.socket()
                s.bind(("127.0.0.1", 0))
                master_port = s.getsockname()[1]
                s.close()
            group_cfg = {
                "world_size": len(devices),
                "master_port": master_port,
                "backend": backend,
            }
        for wid, dev in enumerate(devices):
            parent, child = ctx.Pipe()
            proc = ctx.Process(
                target=_worker_main,
                args=(wid, dev, child, group_cfg, kv_snapshot),
                name=f"rayfed-worker-{wid}",
                daemon=True,
            )
            proc.start()
            child.close()
            self._workers.append(proc)
            self._conns.append(parent)
            self._send_locks.append(threading.Lock())
        self._recv_threads = [
            threading.Thread(target=self._recv_loop, args=(i,), daemon=True)
            for i in range(len(devices))
        ]
        for t in self._recv_threads:
            t.start()
        self._closed = False

    def _recv_loop(self, wid: int):
        conn = self._conns[wid]
        while True:
            try:
                msg = conn.recv_bytes()
            except (EOFError, OSError):
                break
            task_id, is_error, value = cloudpickle.loads(msg)
            with self._fut_lock:
                entry = self._futures.pop(task_id, None)
            if entry is None:
                continue
            fut, _w = entry
            if is_error:
                fut.set_exception(value)
            else:
                fut.set_result(value)
        # Worker gone (crash or shutdown): fail its outstanding futures so
        # callers never hang on a dead process.
        with self._fut_lock:
            dead = [tid for tid, (_f, w) in self._futures.items() if w == wid]
            entries = [self._futures.pop(tid) for tid in dead]
        for fut, _w in entries:
            if not fut.done():
                fut.set_exception(
                    RuntimeError(f"device worker {wid} exited unexpectedly")
                )

    def _send(self, wid: int, op: str, payload) -> ObjectRef:
        task_id = next(self._task_counter)
        fut: Future = Future()
        with self._fut_lock:
            self._futures[task_id] = (fut, wid)
        msg = cloudpickle.dumps((op, task_id, payload))
        with self._send_locks[wid]:
            self._conns[wid].send_bytes(msg)
        return ObjectRef(fut)

    # -- public API -----------------------------------------------------------
    @property
    def num_workers(self) -> int:
        return len(self._workers)

    def submit(self, wid: int, fn, args: tuple = (), kwargs: Optional[dict] = None,
               keep: bool = False) -> ObjectRef:
        return self._send(wid, "task", (fn, args, kwargs or {}, keep))

    def submit_all(self, fn, args: tuple = (), kwargs: Optional[dict] = None,
                   keep: bool = False) -> List[ObjectRef]:
        """Submit the same callable to every worker (collective pattern —
        the callable runs under each worker's rank)."""
        return [
            self.submit(w, fn, args, kwargs, keep) for w in range(self.num_workers)
        ]

    def fetch(self, handle: RemoteHandle) -> ObjectRef:
        return self._send(handle.worker_id, "fetch", handle.obj_id)

    def delete(self, handle: RemoteHandle) -> ObjectRef:
        return self._send(handle.worker_id, "delete", handle.obj_id)

    def shutdown(self, timeout: float = 10.0):
        if self._closed:
            return
        self._closed = True
        for wid in range(len(self._conns)):
            try:
                with self._send_locks[wid]:
                    self._conns[wid].send_bytes(
                        cloudpickle.dumps(("stop", 0, None))
                    )
            except (OSError, BrokenPipeError):
                pass
        for proc in self._workers:
            proc.join(timeout=timeout)
            if proc.is_alive():
                proc.terminate()
        for conn in self._conns:
            conn.close()
