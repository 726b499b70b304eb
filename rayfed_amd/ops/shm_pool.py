"""Same-host shared-memory lane: pooled /dev/shm segments for tensor pushes.

When both parties share the node (the BASELINE configs put all parties on
one 8-GPU MI355X box), shipping multi-GiB tensor bytes through a loopback
socket costs two socket copies at ~2 GB/s.  This lane instead:

  sender:  HIP D2H DMA straight into a pooled SharedMemory segment that was
           ``hipHostRegister``-ed once (so the copy is a real 57 GB/s DMA),
           and puts only ``{segment, offset, nbytes}`` in the frame header;
  receiver: attaches the segment (cached), registers it once, and H2D-DMAs
           directly out of it, CRC-verifying on device.

The receiver acks the frame only after the bytes are consumed, so the
sender can recycle segments on ack.  Falls back transparently to socket
bytes when the peer is remote or shm is disabled (RAYFED_SHM=0).

Trust note: /dev/shm segments are same-user visible; this lane is for
co-located parties (the benchmark topology).  Cross-host traffic always
uses the (optionally TLS) socket path.
"""
from __future__ import annotations

import atexit
import logging
import os
import threading
from multiprocessing import resource_tracker, shared_memory
from typing import Dict, List, Optional

logger = logging.getLogger(__name__)

try:
    import torch
except ImportError:  # pragma: no cover
    torch = None


def shm_enabled() -> bool:
    return os.environ.get("RAYFED_SHM", "1") != "0"


SHM_MIN_BYTES = int(os.environ.get("RAYFED_SHM_MIN_BYTES", str(1 << 20)))


def _unregister_tracker(shm: shared_memory.SharedMemory) -> None:
    """Stop the resource tracker from unlinking attached segments we do not
    own (python's tracker unlinks every segment it saw at process exit)."""
    try:
        resource_tracker.unregister(shm._name, "shared_memory")  # noqa: SLF001
    except Exception:  # noqa: BLE001
        pass


def _maybe_register(ptr: int, size: int) -> bool:
    """hipHostRegister the mapping so D2H/H2D are true DMA; best-effort."""
    if torch is None or not torch.cuda.is_available():
        return False
    try:
        from rayfed_amd.ops import _hip_loader

        _hip_loader.load().host_register(ptr, size)
        return True
    except Exception as e:  # noqa: BLE001
        logger.debug("hipHostRegister failed (%r); shm lane unregistered", e)
        return False


class Segment:
    """An owned pooled segment."""

    def __init__(self, nbytes: int):
        self.shm = shared_memory.SharedMemory(create=True, size=nbytes)
        self.name = self.shm.name
        self.nbytes = nbytes
        import numpy as np

        self.array = np.frombuffer(self.shm.buf, dtype=np.uint8)
        self.torch_view = torch.from_numpy(self.array) if torch is not None else None
        ptr = self.array.ctypes.data
        self.registered = _maybe_register(ptr, nbytes)
        self._ptr = ptr

    def view(self, off: int, nbytes: int) -> memoryview:
        return memoryview(self.array.data)[off : off + nbytes]

    def close(self, unlink: bool = True):
        if self.registered:
            try:
                from rayfed_amd.ops import _hip_loader

                _hip_loader.load().host_unregister(self._ptr)
            except Exception:  # noqa: BLE001
                pass
            self.registered = False
        self.array = None
        self.torch_view = None
        try:
            self.shm.close()
            if unlink:
                self.shm.unlink()
        except Exception:  # noqa: BLE001
            pass


class ShmSegmentPool:
    """Sender-side pool of owned segments (size-bucketed, reused across
    pushes so registration and /dev/shm setup amortize)."""

    def __init__(self, max_segments: int = 8):
        self._free: List[Segment] = []
        self._all: List[Segment] = []
        self._lock = threading.Lock()
        self._max = max_segments
        atexit.register(self.shutdown)

    def acquire(self, nbytes: int) -> Segment:
        with self._lock:
            for i, seg in enumerate(self._free):
                if seg.nbytes >= nbytes:
                    return self._free.pop(i)
        seg = Segment(max(nbytes, SHM_MIN_BYTES))
        with self._lock:
            self._all.append(seg)
        return seg

    def release(self, seg: Segment):
        with self._lock:
            if len(self._free) < self._max:
                self._free.append(seg)
                return
            self._all.remove(seg)
        seg.close(unlink=True)

    def shutdown(self):
        with self._lock:
            segs, self._all, self._free = self._all, [], []
        for seg in segs:
            seg.close(unlink=True)


class AttachedSegment:
    """Receiver-side attachment to a peer-owned segment (cached by name)."""

    def __init__(self, name: str):
        self.shm = shared_memory.SharedMemory(name=name)
        _unregister_tracker(self.shm)
        self.name = name
        import numpy as np

        self.array = np.frombuffer(self.shm.buf, dtype=np.uint8)
        self.torch_view = torch.from_numpy(self.array) if torch is not None else None
        self.registered = _maybe_register(self.array.ctypes.data, len(self.array))
        self._ptr = self.array.ctypes.data
        if os.environ.get("RAYFED_SHM_DEBUG"):
            import sys

            print(
                f"[shm] attached {name}: {len(self.array)>>20} MiB "
                f"registered={self.registered}",
                file=sys.stderr, flush=True,
            )

    def view(self, off: int, nbytes: int) -> memoryview:
        return memoryview(self.array.data)[off : off + nbytes]

    def close(self):
        if self.registered:
            try:
                from rayfed_amd.ops import _hip_loader

                _hip_loader.load().host_unregister(self._ptr)
            except Exception:  # noqa: BLE001
                pass
        self.array = None
        self.torch_view = None
        try:
            self.shm.close()
        except Exception:  # noqa: BLE001
            pass


_send_pool: Optional[ShmSegmentPool] = None
_attach_cache: Dict[str, AttachedSegment] = {}
_lock = threading.Lock()


def get_send_pool() -> ShmSegmentPool:
    global _send_pool
    with _lock:
        if _send_pool is None:
            _send_pool = ShmSegmentPool()
        return _send_pool


def attach(name: str):
    with _lock:
        seg = _attach_cache.get(name)
        if seg is None:
            # Same-process loopback (tests, 1-process demos): reuse the owned
            # segment — hipHostRegister would fail on the already-registered
            # mapping.
            if _send_pool is not None:
                for own in _send_pool._all:  # noqa: SLF001
                    if own.name == name:
                        return own
            seg = AttachedSegment(name)
            _attach_cache[name] = seg
        return seg


def detach_all():
    with _lock:
        segs = list(_attach_cache.values())
        _attach_cache.clear()
    for seg in segs:
        seg.close()


# Close attachments before interpreter teardown so SharedMemory.__del__
# never sees live numpy exports (noisy BufferError otherwise).
atexit.register(detach_all)
