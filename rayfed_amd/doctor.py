"""Environment self-check: ``python -m rayfed_amd.doctor``.

Verifies that a deployment box has everything the fast paths need —
the C++ transport core, the gfx950 HIP extension, dmabuf IPC mode,
loopback bindability — and says which lane each capability maps to.
Exit code 0 when every *required* check passes (GPU-side checks are
informational on CPU-only boxes: the framework runs there on the
asyncio/socket lanes).

New subsystem relative to the reference (it ships no preflight tool);
the checks mirror the lane-selection logic in
``rayfed_amd/ops/tensor_codec.py`` and ``rayfed_amd/proxy/xfer.py``.
"""
from __future__ import annotations

import os
import socket
import sys
from typing import Callable, List, Tuple

# (name, required, check) -> (ok, detail)
Check = Tuple[str, bool, Callable[[], Tuple[bool, str]]]


def _check_python() -> Tuple[bool, str]:
    v = sys.version_info
    ok = v >= (3, 9)
    return ok, f"{v.major}.{v.minor}.{v.micro}"


def _check_torch() -> Tuple[bool, str]:
    try:
        import torch
    except Exception as e:  # noqa: BLE001
        return False, f"import failed: {e}"
    hip = getattr(torch.version, "hip", None)
    return True, f"{torch.__version__} (hip={hip})"


def _check_xfer() -> Tuple[bool, str]:
    from rayfed_amd.proxy.xfer import xfer_available

    if xfer_available():
        return True, "C++ socket core loaded (lane: cpp transport)"
    return False, "_xfer extension missing — falls back to asyncio transport"


def _check_hip_ext() -> Tuple[bool, str]:
    try:
        from rayfed_amd.ops import _hip_loader

        ext = _hip_loader.load()
    except Exception as e:  # noqa: BLE001
        return False, f"HIP extension not loadable ({e}) — device lanes disabled"
    return True, f"gfx950 kernels loaded ({type(ext).__name__})"


def _check_gpu() -> Tuple[bool, str]:
    import torch

    if not torch.cuda.is_available():
        return False, "no GPU visible (CPU-only mode: socket lanes only)"
    name = torch.cuda.get_device_name(0)
    n = torch.cuda.device_count()
    return True, f"{n}x {name}"


def _check_ipc_env() -> Tuple[bool, str]:
    v = os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY")
    if v == "0":
        return True, "HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC)"
    return (
        False,
        f"HSA_ENABLE_IPC_MODE_LEGACY={v!r} — device-IPC lane needs 0 on "
        "dmabuf-only hosts (hipIpcGetMemHandle fails otherwise)",
    )


def _check_loopback() -> Tuple[bool, str]:
    try:
        with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        return True, f"bindable (sample port {port})"
    except OSError as e:
        return False, str(e)


def _check_serialization() -> Tuple[bool, str]:
    from rayfed_amd._private import serialization

    blob = serialization.dumps({"k": [1, 2, 3]})
    ok = serialization.loads(blob) == {"k": [1, 2, 3]}
    return ok, f"roundtrip ok ({len(blob)} B probe)"


def run_checks() -> List[Tuple[str, bool, bool, str]]:
    """Run all checks; returns [(name, required, ok, detail)]."""
    checks: List[Check] = [
        ("python", True, _check_python),
        ("torch", True, _check_torch),
        ("serialization", True, _check_serialization),
        ("loopback", True, _check_loopback),
        ("cpp transport", False, _check_xfer),
        ("hip kernels", False, _check_hip_ext),
        ("gpu", False, _check_gpu),
        ("ipc env", False, _check_ipc_env),
    ]
    out = []
    for name, required, fn in checks:
        try:
            ok, detail = fn()
        except Exception as e:  # noqa: BLE001 — a crashed probe is a failure
            ok, detail = False, f"check crashed: {e}"
        out.append((name, required, ok, detail))
    return out


def main(argv=None) -> int:
    results = run_checks()
    width = max(len(n) for n, *_ in results)
    failed_required = False
    for name, required, ok, detail in results:
        mark = "ok  " if ok else ("FAIL" if required else "warn")
        print(f"[{mark}] {name.ljust(width)}  {detail}")
        if required and not ok:
            failed_required = True
    if failed_required:
        print("doctor: required checks failed", file=sys.stderr)
        return 1
    print("doctor: environment ok (warnings above, if any, disable fast lanes only)")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
