"""In-process internal KV store.

The reference persists cluster/job config in Ray's GCS internal KV so that
proxy *actor processes* can read it back
(/root/reference/fed/_private/compatible_utils.py:68-188).  This engine hosts
the proxies in the driver process, so the KV is a process-local store — but it
keeps the same API (``initialize/put/get/delete/reset``) and the same
job-prefixed key scheme ``RAYFED#{job_name}#{key}`` so config plumbing and the
KV tests carry over unchanged.
"""
from __future__ import annotations

import threading
from typing import Dict, Optional

_store: Dict[bytes, bytes] = {}
_store_lock = threading.Lock()

kv = None  # module-level singleton, mirrors compatible_utils.kv


class InternalKv:
    """Process-local KV with job-name key prefixing."""

    def __init__(self, job_name: str):
        self._job_name = job_name

    def _prefix(self, key) -> bytes:
        if isinstance(key, str):
            key = key.encode()
        return f"RAYFED#{self._job_name}#".encode() + key

    def initialize(self) -> bool:
        return True

    def put(self, key, value) -> bool:
        if isinstance(value, str):
            value = value.encode()
        with _store_lock:
            _store[self._prefix(key)] = value
        return True

    def get(self, key) -> Optional[bytes]:
        with _store_lock:
            return _store.get(self._prefix(key))

    def delete(self, key) -> bool:
        with _store_lock:
            _store.pop(self._prefix(key), None)
        return True

    def reset(self) -> bool:
        """Drop every key belonging to this job."""
        prefix = f"RAYFED#{self._job_name}#".encode()
        with _store_lock:
            for k in [k for k in _store if k.startswith(prefix)]:
                del _store[k]
        return True


def _init_internal_kv(job_name: str) -> InternalKv:
    global kv
    if kv is None:
        kv = InternalKv(job_name)
        kv.initialize()
    return kv


def _clear_internal_kv() -> None:
    global kv
    if kv is not None:
        kv.delete("CLUSTER_CONFIG")
        kv.delete("JOB_CONFIG")
        kv.reset()
        kv = None
