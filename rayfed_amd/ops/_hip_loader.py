"""Loader for the in-tree HIP extension (built by setup.py / __graft_entry__.build).

The extension is compiled for gfx950 only and lives IN-TREE
(``rayfed_amd/_hip.<abi>.so``) so it travels with the repo snapshot to GPU
boxes.  On a machine with a visible HIP device the extension is mandatory —
``load()`` raises ImportError rather than letting callers fall back to a
silent eager path.
"""
from __future__ import annotations

import importlib
import threading

_ext = None
_lock = threading.Lock()


def load():
    global _ext
    if _ext is not None:
        return _ext
    with _lock:
        if _ext is None:
            try:
                import torch  # noqa: F401 — loads libc10/libtorch first

                _ext = importlib.import_module("rayfed_amd._hip")
            except ImportError as e:
                raise ImportError(
                    "rayfed_amd HIP extension not built. Run "
                    "`python setup.py build_ext --inplace` (or "
                    "__graft_entry__.build()) with PYTORCH_ROCM_ARCH=gfx950. "
                    f"Original error: {e}"
                ) from e
        return _ext


def is_available() -> bool:
    try:
        load()
        return True
    except ImportError:
        return False
