"""Cross-silo wire format: raw-bytes frames over generic gRPC methods.

The reference ships a one-RPC protobuf service
(/root/reference/fed/grpc/fed.proto:5-19 — ``SendData(bytes data, string
upstream_seq_id, string downstream_seq_id, string job_name)``).  This engine
speaks the same *logical* protocol but frames it by hand over gRPC's
generic-handler API with identity (de)serializers:

- no protobuf codegen dependency;
- the (potentially multi-GiB) tensor payload is never copied into a protobuf
  message — the frame is ``header || payload`` and the payload slice is a
  zero-copy ``memoryview`` on the receive side;
- large bodies stream as parallel STRIPE sub-frames at the transport layer
  (csrc/xfer_core.cpp) — no frame-level streaming variant is needed.

Frame layout (little-endian)::

    magic   4 B  = b"RFED"
    version 1 B  = 1
    kind    1 B  (0 = cloudpickle object, 1 = tensor manifest, 2 = error)
    flags   2 B  (reserved)
    hlen    4 B  header length
    header  hlen B   msgpack dict {job, up, down, ...}
    payload rest

Responses are msgpack ``{code: int, result: str}`` with HTTP-style codes
(200 OK, 417 job-name mismatch — parity with grpc_proxy.py:310-320).
"""
from __future__ import annotations

import struct
from typing import Any, Dict, Tuple, Union

import msgpack

MAGIC = b"RFED"
VERSION = 1

KIND_PICKLE = 0
KIND_TENSOR = 1
KIND_ERROR = 2
# Chunk-streamed tensor frame (C++ transport only): the main frame's payload
# is the INNER tensor frame's prefix; the payload bytes travel as xk sidecar
# frames under derived seq ids, consumed progressively (H2D overlaps network
# arrival — tensor_codec.decode_streamed).
KIND_CHUNKED = 3

_PREFIX = struct.Struct("<4sBBHI")

SERVICE_NAME = "rayfedamd.GrpcService"
SEND_DATA_METHOD = f"/{SERVICE_NAME}/SendData"


def encode_frame(
    kind: int,
    header: Dict[str, Any],
    payload: Union[bytes, memoryview, bytearray] = b"",
) -> bytes:
    hdr = msgpack.packb(header, use_bin_type=True)
    return _PREFIX.pack(MAGIC, VERSION, kind, 0, len(hdr)) + hdr + bytes(payload)


def encode_frame_prefix(kind: int, header: Dict[str, Any]) -> bytes:
    """Frame prefix only — the payload parts are written separately by the
    transport (zero-copy streaming of pinned staging views)."""
    hdr = msgpack.packb(header, use_bin_type=True)
    return _PREFIX.pack(MAGIC, VERSION, kind, 0, len(hdr)) + hdr


def decode_frame(data: bytes) -> Tuple[int, Dict[str, Any], memoryview]:
    """Return (kind, header, payload-view).  The payload is a zero-copy view
    into the request buffer."""
    if len(data) < _PREFIX.size:
        raise ValueError("short frame")
    magic, version, kind, _flags, hlen = _PREFIX.unpack_from(data, 0)
    if magic != MAGIC:
        raise ValueError("bad frame magic")
    if version != VERSION:
        raise ValueError(f"unsupported frame version {version}")
    off = _PREFIX.size
    header = msgpack.unpackb(data[off : off + hlen], raw=False)
    return kind, header, memoryview(data)[off + hlen :]


def encode_response(code: int, result: str = "") -> bytes:
    return msgpack.packb({"code": code, "result": result}, use_bin_type=True)


def decode_response(data: bytes) -> Dict[str, Any]:
    return msgpack.unpackb(data, raw=False)


def identity_serializer(x: bytes) -> bytes:
    return x


def identity_deserializer(x: bytes) -> bytes:
    return x
