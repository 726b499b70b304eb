"""Lightweight event tracing → Chrome trace format.

The reference has no tracing at all (only per-proxy op
counters).  This engine records cross-silo and task events into an
in-memory buffer and writes a ``chrome://tracing`` / Perfetto-compatible
JSON at shutdown.

Enable with ``fed.init(config={"trace_file": "/tmp/fed_trace.json"})`` or
``RAYFED_TRACE=/tmp/fed_trace.json``.  Overhead when disabled: one module
bool check per instrumentation site.
"""
from __future__ import annotations

import json
import os
import threading
import time
from contextlib import contextmanager
from typing import Any, Dict, List, Optional

enabled = False
_events: List[Dict[str, Any]] = []
_lock = threading.Lock()
_path: Optional[str] = None
_t0 = time.perf_counter()


def _now_us() -> float:
    return (time.perf_counter() - _t0) * 1e6


def configure(path: Optional[str]) -> None:
    """Turn tracing on (path != None) or off; called from fed.init."""
    global enabled, _path
    _path = path or os.environ.get("RAYFED_TRACE") or None
    enabled = _path is not None
    if enabled:
        with _lock:
            _events.clear()


def event(name: str, cat: str, **args: Any) -> None:
    """Instant event."""
    if not enabled:
        return
    with _lock:
        _events.append(
            {
                "name": name,
                "cat": cat,
                "ph": "i",
                "s": "t",
                "ts": _now_us(),
                "pid": os.getpid(),
                "tid": threading.get_ident() & 0xFFFF,
                "args": args,
            }
        )


@contextmanager
def span(name: str, cat: str, **args: Any):
    """Complete-event span around a block."""
    if not enabled:
        yield
        return
    ts = _now_us()
    try:
        yield
    finally:
        with _lock:
            _events.append(
                {
                    "name": name,
                    "cat": cat,
                    "ph": "X",
                    "ts": ts,
                    "dur": _now_us() - ts,
                    "pid": os.getpid(),
                    "tid": threading.get_ident() & 0xFFFF,
                    "args": args,
                }
            )


def flush() -> Optional[str]:
    """Write the trace file (appending a per-pid suffix so every party in a
    multi-process job gets its own file).  Returns the path written."""
    if not enabled or _path is None:
        return None
    out = _path
    root, ext = os.path.splitext(_path)
    out = f"{root}.{os.getpid()}{ext or '.json'}"
    with _lock:
        payload = {"traceEvents": list(_events)}
    try:
        with open(out, "w") as f:
            json.dump(payload, f)
        return out
    except OSError:
        return None
