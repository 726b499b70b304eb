"""GPU data plane: device tensor ⇄ host staging for the cross-silo path.

MI355X-native replacement for the reference's CPU pickle of payloads
(/root/reference/fed/proxy/grpc/grpc_proxy.py:202).  Pipeline per tensor
(SURVEY.md §2.3 / §7 step 3):

  send:  HIP pack+CRC32 kernel on a side stream → chunked hipMemcpyAsync
         D2H into pinned staging → bytes handed to the gRPC frame
  recv:  frame bytes → pinned staging → hipMemcpyAsync H2D on a side
         stream → HIP CRC32 verify kernel → device tensor

The CRC32 and pack kernels live in the in-tree HIP extension
(``csrc/pack_crc.hip`` → ``rayfed_amd._hip``); when the extension or a GPU is
absent this module is not instantiated and the CPU fallback in
``tensor_codec`` applies.  On a GPU box the HIP extension is REQUIRED — we
fail loudly rather than fall back silently (so a "GPU test" can never pass on
an eager CPU path by accident).
"""
from __future__ import annotations

import logging
import threading
from typing import List, Optional, Tuple

from rayfed_amd.config import GpuDataPlaneConfig

logger = logging.getLogger(__name__)

try:
    import torch
except ImportError:  # pragma: no cover
    torch = None


def _load_hip_ext():
    """Import the in-tree HIP extension; raise ImportError when missing."""
    from rayfed_amd.ops import _hip_loader

    return _hip_loader.load()


class GpuDataPlane:
    """Per-process staging machinery for one visible HIP device."""

    def __init__(self, config: GpuDataPlaneConfig, device: Optional[int] = None):
        if torch is None or not torch.cuda.is_available():
            raise RuntimeError("GpuDataPlane requires a visible HIP device")
        self.config = config
        self.device = torch.device("cuda", device or torch.cuda.current_device())
        # HIP extension is mandatory on-GPU: loud failure over silent fallback.
        self._ext = _load_hip_ext()
        # Side streams so pack/copy overlap compute and each other.
        self._d2h_stream = torch.cuda.Stream(device=self.device)
        self._h2d_stream = torch.cuda.Stream(device=self.device)
        self._lock = threading.Lock()
        self._pinned: List[torch.Tensor] = []

    # -- pinned staging pool --------------------------------------------------
    def _get_pinned(self, nbytes: int) -> torch.Tensor:
        with self._lock:
            for i, buf in enumerate(self._pinned):
                if buf.numel() >= nbytes:
                    return self._pinned.pop(i)
        return torch.empty(
            max(nbytes, self.config.chunk_bytes), dtype=torch.uint8, pin_memory=True
        )

    def _put_pinned(self, buf: torch.Tensor) -> None:
        with self._lock:
            if len(self._pinned) < self.config.staging_buffers:
                self._pinned.append(buf)

    # -- send path ------------------------------------------------------------
    def pack_to_host(self, t: "torch.Tensor") -> Tuple[memoryview, Optional[int]]:
        """Flatten ``t`` to raw bytes in pinned host memory; return
        (bytes view, crc32|None).  CRC is computed on-device by the HIP
        kernel, overlapped with the D2H copy on the side stream."""
        t = t.detach()
        if not t.is_contiguous():
            t = t.contiguous()
        nbytes = t.numel() * t.element_size()
        flat = t.view(-1).view(torch.uint8) if nbytes else t.new_empty(0, dtype=torch.uint8)
        staging = self._get_pinned(nbytes)
        crc = None
        with torch.cuda.stream(self._d2h_stream):
            self._d2h_stream.wait_stream(torch.cuda.current_stream(self.device))
            if nbytes:
                staging[:nbytes].copy_(flat, non_blocking=True)
            if self.config.verify_crc and nbytes:
                crc_t = self._ext.crc32(flat)
            done = torch.cuda.Event()
            done.record(self._d2h_stream)
        done.synchronize()
        if self.config.verify_crc and nbytes:
            crc = int(crc_t.item()) & 0xFFFFFFFF
        out = staging[:nbytes].numpy().data
        # NOTE: the staging buffer is handed to the frame encoder as a view;
        # it returns to the pool only after the bytes are copied onto the
        # wire (frames.encode_frame materializes with bytes()).
        self._put_pinned(staging)
        return out, crc

    # -- recv path ------------------------------------------------------------
    def unpack_from_host(
        self,
        raw: memoryview,
        dtype: "torch.dtype",
        shape: List[int],
        crc_expect: Optional[int],
    ) -> "torch.Tensor":
        nbytes = len(raw)
        staging = self._get_pinned(nbytes)
        if nbytes:
            staging[:nbytes].copy_(
                torch.frombuffer(bytearray(raw), dtype=torch.uint8)
            )
        out = torch.empty(shape, dtype=dtype, device=self.device)
        with torch.cuda.stream(self._h2d_stream):
            flat = out.view(-1).view(torch.uint8) if nbytes else None
            if nbytes:
                flat.copy_(staging[:nbytes], non_blocking=True)
                if self.config.verify_crc and crc_expect is not None:
                    crc_t = self._ext.crc32(flat)
            done = torch.cuda.Event()
            done.record(self._h2d_stream)
        done.synchronize()
        if nbytes and self.config.verify_crc and crc_expect is not None:
            crc = int(crc_t.item()) & 0xFFFFFFFF
            if crc != crc_expect:
                raise ValueError(
                    f"GPU tensor CRC mismatch: expected {crc_expect:#x}, got {crc:#x}"
                )
        self._put_pinned(staging)
        return out


_plane: Optional[GpuDataPlane] = None
_plane_lock = threading.Lock()


def maybe_create_gpu_plane(config_dict: Optional[dict] = None) -> Optional[GpuDataPlane]:
    """Create (once) the process's GPU data plane if a HIP device is visible."""
    global _plane
    if torch is None or not torch.cuda.is_available():
        return None
    with _plane_lock:
        if _plane is None:
            _plane = GpuDataPlane(GpuDataPlaneConfig.from_dict(config_dict))
        return _plane
