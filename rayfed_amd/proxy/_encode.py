"""Shared request encoding for sender proxies (TCP and gRPC)."""
from __future__ import annotations

import asyncio
from typing import Optional

from rayfed_amd._private import serialization
from rayfed_amd.exceptions import FedRemoteError
from rayfed_amd.ops import tensor_codec
from rayfed_amd.proxy.grpc import frames

# Below this (estimated) payload size, encode inline on the I/O loop — the
# executor hop costs more than the pickle.
_INLINE_HINT = object()


async def encode_request(
    job_name: str,
    data,
    upstream_seq_id,
    downstream_seq_id,
    gpu_plane=None,
    extra_header: Optional[dict] = None,
) -> bytes:
    header = {
        "job": job_name,
        "up": str(upstream_seq_id),
        "down": str(downstream_seq_id),
    }
    if extra_header:
        header.update(extra_header)
    if isinstance(data, FedRemoteError):
        return frames.encode_frame(
            frames.KIND_ERROR, header, serialization.dumps(data)
        )

    # Cheap scalar/bytes payloads encode inline (no executor hop); anything
    # else — containers, tensors, user objects — encodes in the pool so a
    # multi-GiB pickle or a GPU pack never blocks the I/O loop.
    if isinstance(data, (int, float, str, bytes, bool, type(None))) and (
        not isinstance(data, (str, bytes)) or len(data) < 64 * 1024
    ):
        extras, parts = tensor_codec.encode(data, gpu_plane)
    else:
        loop = asyncio.get_running_loop()
        extras, parts = await loop.run_in_executor(
            None, tensor_codec.encode, data, gpu_plane
        )
    if extras["tensors"]:
        header.update(extras)
        return frames.encode_frame(
            frames.KIND_TENSOR, header, b"".join(bytes(p) for p in parts)
        )
    return frames.encode_frame(frames.KIND_PICKLE, header, parts[0])
