"""ObjectRef — the in-party future handle.

Moral equivalent of Ray's ``ObjectRef`` but backed by a
``concurrent.futures.Future`` inside the party driver (or a handle into a GPU
worker process).  Resolution is event-driven: ``result()`` blocks on the
future; there are no poll loops anywhere on this path.
"""
from __future__ import annotations

import itertools
from concurrent.futures import Future
from typing import Any, Optional

_ref_counter = itertools.count()


class ObjectRef:
    """A handle to the (eventual) result of a task in the local party."""

    __slots__ = ("_future", "_id")

    def __init__(self, future: Future, ref_id: Optional[int] = None):
        self._future = future
        self._id = ref_id if ref_id is not None else next(_ref_counter)

    # -- construction helpers -------------------------------------------------
    @classmethod
    def from_value(cls, value: Any) -> "ObjectRef":
        fut: Future = Future()
        fut.set_result(value)
        return cls(fut)

    @classmethod
    def from_exception(cls, exc: BaseException) -> "ObjectRef":
        fut: Future = Future()
        fut.set_exception(exc)
        return cls(fut)

    def chain(self, fn) -> "ObjectRef":
        """A new ref whose value is ``fn(self.result())`` (lazy, callback-driven)."""
        out: Future = Future()

        def _done(f: Future):
            try:
                out.set_result(fn(f.result()))
            except BaseException as e:  # noqa: BLE001 - propagate task errors
                out.set_exception(e)

        self._future.add_done_callback(_done)
        return ObjectRef(out)

    # -- protocol -------------------------------------------------------------
    @property
    def future(self) -> Future:
        return self._future

    def result(self, timeout: Optional[float] = None) -> Any:
        return self._future.result(timeout)

    def done(self) -> bool:
        return self._future.done()

    def id(self) -> int:
        return self._id

    def __repr__(self) -> str:
        state = "done" if self._future.done() else "pending"
        return f"ObjectRef({self._id}, {state})"
