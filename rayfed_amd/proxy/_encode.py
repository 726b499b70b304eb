"""Shared request encoding for sender proxies (TCP and gRPC)."""
from __future__ import annotations

import asyncio
from typing import List, Optional

from rayfed_amd._private import serialization
from rayfed_amd.exceptions import FedRemoteError
from rayfed_amd.ops import tensor_codec
from rayfed_amd.proxy.grpc import frames


class EncodedRequest:
    """A framed request as (prefix, payload parts): the TCP transport writes
    parts sequentially (no join copy — GPU payload parts are views over
    pinned staging); :meth:`release` returns pooled buffers after the ack."""

    __slots__ = ("prefix", "parts", "_extras")

    def __init__(self, prefix: bytes, parts: List, extras: Optional[dict] = None):
        self.prefix = prefix
        self.parts = parts
        self._extras = extras

    @property
    def total_len(self) -> int:
        return len(self.prefix) + sum(len(p) for p in self.parts)

    def to_bytes(self) -> bytes:
        return self.prefix + b"".join(bytes(p) for p in self.parts)

    def release(self) -> None:
        if self._extras is not None:
            tensor_codec.release_parts(self._extras)


async def encode_request(
    job_name: str,
    data,
    upstream_seq_id,
    downstream_seq_id,
    gpu_plane=None,
    extra_header: Optional[dict] = None,
    shm: bool = False,
) -> EncodedRequest:
    header = {
        "job": job_name,
        "up": str(upstream_seq_id),
        "down": str(downstream_seq_id),
    }
    if extra_header:
        header.update(extra_header)
    if isinstance(data, FedRemoteError):
        payload = serialization.dumps(data)
        return EncodedRequest(
            frames.encode_frame_prefix(frames.KIND_ERROR, header), [payload]
        )

    # Cheap scalar/bytes payloads encode inline (no executor hop); anything
    # else — containers, tensors, user objects — encodes in the pool so a
    # multi-GiB pickle or a GPU pack never blocks the I/O loop.
    if isinstance(data, (int, float, bool, type(None))) or (
        isinstance(data, (str, bytes)) and len(data) < 64 * 1024
    ):
        extras, parts = tensor_codec.encode(data, gpu_plane, shm)
    else:
        loop = asyncio.get_running_loop()
        extras, parts = await loop.run_in_executor(
            None, tensor_codec.encode, data, gpu_plane, shm
        )
    if extras["tensors"]:
        releases_extras = extras if "_releases" in extras else None
        wire_header = {k: v for k, v in extras.items() if k != "_releases"}
        header.update(wire_header)
        return EncodedRequest(
            frames.encode_frame_prefix(frames.KIND_TENSOR, header),
            parts,
            releases_extras,
        )
    return EncodedRequest(
        frames.encode_frame_prefix(frames.KIND_PICKLE, header), [parts[0]]
    )
