"""Fed actor handles — party-placed actors with ordered method execution.

Parity: /root/reference/fed/_private/fed_actor.py:26-145.  A FedActorHandle
fabricates a FedActorMethod per attribute access (validated against the class
body); the real actor exists only in the owning party, backed by the in-party
executor's single-threaded ActorHandle (same ordering contract as a Ray
actor).
"""
from __future__ import annotations

import logging
from typing import Dict, Optional

from rayfed_amd._private.call_holder import FedCallHolder
from rayfed_amd.runtime.executor import ActorHandle, Executor
from rayfed_amd.utils import materialize

logger = logging.getLogger(__name__)


def _invoke_materialized(fn, args, kwargs):
    from rayfed_amd._private import tracing

    if tracing.enabled:
        with tracing.span(getattr(fn, "__name__", "task"), "task"):
            args, kwargs = materialize((args, kwargs))
            return fn(*args, **kwargs)
    args, kwargs = materialize((args, kwargs))
    return fn(*args, **kwargs)


def _invoke_on_device(executor, device, fn, args, kwargs):
    """Runs on a driver thread: materialize ObjectRef args locally, then
    execute the call in GPU worker `device` and wait its result."""
    args, kwargs = materialize((args, kwargs))
    return executor.submit_on_device(device, fn, args, kwargs).result()


# -- worker-side actor helpers (run inside the GPU worker process) ------------
def _worker_create_actor(cls, args, kwargs):
    return cls(*args, **kwargs)


def _worker_call_method(instance, method_name, args, kwargs):
    return getattr(instance, method_name)(*args, **kwargs)


class FedActorHandle:
    def __init__(
        self,
        fed_class_task_id: int,
        addresses: Dict,
        cls,
        party: str,
        node_party: str,
        options: Optional[Dict],
        executor: Executor,
    ) -> None:
        self._fed_class_task_id = fed_class_task_id
        self._addresses = addresses
        self._body = cls
        self._party = party
        self._node_party = node_party
        self._options = options or {}
        self._executor = executor
        self._actor_handle: Optional[ActorHandle] = None

    def __getattr__(self, method_name: str):
        # `__getattr__` runs only for *missing* attributes; every public
        # method name resolves to a FedActorMethod.
        if method_name.startswith("_"):
            raise AttributeError(method_name)
        if not hasattr(self._body, method_name):
            raise AttributeError(
                f"{self._body.__name__} has no method {method_name!r}"
            )
        call_holder = FedCallHolder(
            self._node_party,
            lambda args, kwargs, _m=method_name: self._execute_remote_method(
                _m, self._options, args, kwargs
            ),
        )
        return FedActorMethod(self, method_name, call_holder)

    def _execute_impl(self, cls_args, cls_kwargs):
        """Create the real actor — only in the owning party
        (reference fed_actor.py:78-91).  With ``options(device=k)`` the
        actor lives in GPU worker process k, its state device-resident."""
        if self._node_party != self._party:
            return None
        device = self._options.get("device") if self._options else None
        if device is not None:
            self._device = device
            # Serialize device-actor calls via a 1-thread local actor that
            # proxies into the worker (ordering parity with Ray actors).
            self._actor_handle = self._executor.create_actor(
                _DeviceActorShellFactory(
                    self._executor, device, self._body
                ),
                args=(cls_args, cls_kwargs),
                name=f"{self._body.__name__}-{self._fed_class_task_id}-dev{device}",
            )
            return self._actor_handle.ready_ref
        self._actor_handle = self._executor.create_actor(
            _MaterializingActorFactory(self._body),
            args=(cls_args, cls_kwargs),
            name=f"{self._body.__name__}-{self._fed_class_task_id}",
        )
        return self._actor_handle.ready_ref

    def _execute_remote_method(self, method_name, options, args, kwargs):
        if self._actor_handle is None:
            raise RuntimeError(
                f"actor {self._body.__name__} was not created in this party"
            )
        num_returns = 1
        if options and "num_returns" in options:
            num_returns = options["num_returns"]
        return self._actor_handle.call(
            "call_method",
            args=(method_name, args, kwargs),
            num_returns=num_returns,
        )

    def _kill(self):
        if self._actor_handle is not None:
            self._actor_handle.kill()


class _MaterializingActorFactory:
    """Wraps the user class so __init__ args (possibly ObjectRefs / recv
    barriers) materialize inside the actor thread, and methods are invoked
    through one dispatch point with the same materialization."""

    __name__ = "_MaterializingActorFactory"

    def __init__(self, body):
        self._body = body

    def __call__(self, cls_args, cls_kwargs):
        args, kwargs = materialize((cls_args, cls_kwargs))
        return _ActorShell(self._body(*args, **kwargs))


class _ActorShell:
    def __init__(self, instance):
        self._instance = instance

    def call_method(self, method_name, args, kwargs):
        args, kwargs = materialize((args, kwargs))
        return getattr(self._instance, method_name)(*args, **kwargs)


class _DeviceActorShellFactory:
    """Creates the actor INSIDE GPU worker `device`; the driver-side shell
    holds only a RemoteHandle and proxies method calls."""

    __name__ = "_DeviceActorShellFactory"

    def __init__(self, executor, device, body):
        self._executor = executor
        self._device = device
        self._body = body

    def __call__(self, cls_args, cls_kwargs):
        args, kwargs = materialize((cls_args, cls_kwargs))
        handle = self._executor.submit_on_device(
            self._device, _worker_create_actor, (self._body, args, kwargs),
            keep=True,
        ).result()
        return _DeviceActorShell(self._executor, self._device, handle)


class _DeviceActorShell:
    def __init__(self, executor, device, remote_handle):
        self._executor = executor
        self._device = device
        self._handle = remote_handle

    def call_method(self, method_name, args, kwargs):
        args, kwargs = materialize((args, kwargs))
        return self._executor.submit_on_device(
            self._device,
            _worker_call_method,
            (self._handle, method_name, args, kwargs),
        ).result()


class FedActorMethod:
    def __init__(self, handle: FedActorHandle, method_name: str, call_holder: FedCallHolder):
        self._handle = handle
        self._method_name = method_name
        self._call_holder = call_holder

    def remote(self, *args, **kwargs):
        return self._call_holder.internal_remote(*args, **kwargs)

    def options(self, **options):
        self._call_holder.options(**options)
        return self
