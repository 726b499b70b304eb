"""In-party task/actor executor.

Replaces Ray's task/actor substrate for one party's driver
process.  Control-plane tasks run on a shared thread pool; actors get a
dedicated single-thread executor so their method calls serialize in submission
order — the same ordering contract Ray actors give the reference
(/root/reference/fed/_private/fed_actor.py:78-112 relies on it).

GPU data-plane tasks can be routed to per-device worker processes via the
``device`` option (see ``rayfed_amd.runtime.worker``); the default is
driver-local execution, which on one MI355X node with the GIL released inside
torch/HIP calls is both lower-latency and simpler than Ray's process fan-out.
"""
from __future__ import annotations

import logging
import queue
import threading
from concurrent.futures import Future, ThreadPoolExecutor
from typing import Any, Dict, List, Optional, Sequence, Union

from rayfed_amd.runtime.object_ref import ObjectRef

logger = logging.getLogger(__name__)


def _split_returns(ref: ObjectRef, num_returns: int) -> List[ObjectRef]:
    """Split a task returning ``num_returns`` values into per-value refs."""

    def _index(i: int):
        def pick(value):
            if not isinstance(value, (tuple, list)) or len(value) != num_returns:
                raise ValueError(
                    f"task declared num_returns={num_returns} but returned "
                    f"{type(value).__name__}"
                )
            return value[i]

        return pick

    return [ref.chain(_index(i)) for i in range(num_returns)]


class Executor:
    """Shared thread-pool executor for a party driver."""

    def __init__(self, max_workers: int = 16):
        self._pool = ThreadPoolExecutor(
            max_workers=max_workers, thread_name_prefix="rayfed-task"
        )
        self._lock = threading.Lock()
        self._actors: List["ActorHandle"] = []
        self._shutdown = False
        # Optional per-GPU worker processes (rayfed_amd.runtime.worker):
        # tasks submitted with device=k route to worker k and their results
        # can stay device-resident.
        self._worker_pool = None

    def attach_worker_pool(self, pool) -> None:
        self._worker_pool = pool

    @property
    def worker_pool(self):
        return self._worker_pool

    def submit_on_device(self, device: int, fn, args=(), kwargs=None,
                         keep: bool = False):
        if self._worker_pool is None:
            raise RuntimeError(
                "no device worker pool attached; pass party_gpus to fed.init"
            )
        return self._worker_pool.submit(device, fn, tuple(args), kwargs, keep)

    # -- tasks ----------------------------------------------------------------
    def submit(
        self,
        fn,
        args: Sequence[Any] = (),
        kwargs: Optional[Dict[str, Any]] = None,
        num_returns: int = 1,
    ) -> Union[ObjectRef, List[ObjectRef]]:
        kwargs = kwargs or {}
        fut = self._pool.submit(fn, *args, **kwargs)
        ref = ObjectRef(fut)
        if num_returns == 1:
            return ref
        return _split_returns(ref, num_returns)

    # -- actors ---------------------------------------------------------------
    def create_actor(
        self,
        cls,
        args: Sequence[Any] = (),
        kwargs: Optional[Dict[str, Any]] = None,
        name: Optional[str] = None,
    ) -> "ActorHandle":
        handle = ActorHandle(cls, args, kwargs or {}, name=name)
        with self._lock:
            self._actors.append(handle)
        return handle

    # -- lifecycle ------------------------------------------------------------
    def shutdown(self, wait: bool = True) -> None:
        with self._lock:
            if self._shutdown:
                return
            self._shutdown = True
            actors = list(self._actors)
            self._actors.clear()
        for a in actors:
            a.kill(no_restart=True)
        if self._worker_pool is not None:
            self._worker_pool.shutdown()
            self._worker_pool = None
        self._pool.shutdown(wait=wait)


_ACTOR_STOP = object()


class ActorHandle:
    """A single-threaded actor: methods execute strictly in submission order.

    A dedicated worker loop (queue.Queue + bare Futures) instead of a
    1-thread ThreadPoolExecutor: the pool's per-submit work-item machinery
    measured ~10 us per call on the tiny-task hot path."""

    def __init__(self, cls, args, kwargs, name: Optional[str] = None):
        self._cls = cls
        self._name = name or f"{cls.__name__}-actor"
        self._q: "queue.Queue" = queue.Queue()
        self._killed = False
        self._lock = threading.Lock()
        # Instantiate asynchronously, like Ray's deferred actor creation:
        # creation errors surface on the first method call's ref.
        self._instance_fut: Future = Future()
        self._q.put((self._instance_fut, lambda: cls(*args, **kwargs)))
        self._ready_ref = ObjectRef(self._instance_fut)
        self._thread = threading.Thread(
            target=self._loop, name=f"rayfed-actor-{self._name}", daemon=True
        )
        self._thread.start()

    def _loop(self):
        while True:
            item = self._q.get()
            if item is _ACTOR_STOP:
                break
            fut, fn = item
            if not fut.set_running_or_notify_cancel():
                continue
            try:
                fut.set_result(fn())
            except BaseException as e:  # noqa: BLE001 — task errors go to the ref
                fut.set_exception(e)

    @property
    def ready_ref(self) -> ObjectRef:
        return self._ready_ref

    def call(
        self,
        method_name: str,
        args: Sequence[Any] = (),
        kwargs: Optional[Dict[str, Any]] = None,
        num_returns: int = 1,
    ) -> Union[ObjectRef, List[ObjectRef]]:
        kwargs = kwargs or {}
        with self._lock:
            if self._killed:
                return ObjectRef.from_exception(
                    RuntimeError(f"actor {self._name} has been killed")
                )

            def _invoke():
                instance = self._instance_fut.result()
                method = getattr(instance, method_name)
                return method(*args, **kwargs)

            fut: Future = Future()
            self._q.put((fut, _invoke))
        ref = ObjectRef(fut)
        if num_returns == 1:
            return ref
        return _split_returns(ref, num_returns)

    def kill(self, no_restart: bool = True) -> None:
        with self._lock:
            if self._killed:
                return
            self._killed = True
        # Cancel whatever is still queued, then stop the worker.
        try:
            while True:
                item = self._q.get_nowait()
                if item is not _ACTOR_STOP:
                    item[0].cancel()
        except queue.Empty:
            pass
        self._q.put(_ACTOR_STOP)

    def __repr__(self) -> str:
        return f"ActorHandle({self._name})"
