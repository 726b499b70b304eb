"""Shared receive mailbox: seq-id-keyed futures + payload deserialization.

Used by every receiver proxy (TCP and gRPC).  Equivalent of the reference's
two 2-dim dicts + asyncio.Events (grpc_proxy.py:241-243,328-341), collapsed
into one ``{(up, down): Future}`` map: whichever side arrives first creates
the future; delivery resolves it; the reader pops it after consumption.
"""
from __future__ import annotations

import asyncio
from typing import Dict, Optional, Tuple

from rayfed_amd._private import constants, serialization
from rayfed_amd.ops import tensor_codec
from rayfed_amd.proxy.grpc import frames

# Payloads at or above this size deserialize in the thread pool so the I/O
# loop stays responsive.
OFFLOAD_BYTES = 256 * 1024


class Mailbox:
    def __init__(self, job_name: str, allowed_list: Optional[dict] = None):
        self._job_name = job_name
        self._allowed_list = allowed_list
        self._slots: Dict[Tuple[str, str], asyncio.Future] = {}
        self.received_op_count = 0
        self.gpu_plane = None

    _KIND_OBJ = -1  # already-decoded object (shm lane eager decode)

    def check_job(self, header: dict) -> Optional[Tuple[int, str]]:
        job_name = header.get("job", "")
        if job_name != self._job_name:
            return (
                417,
                f"JobName mis-match: expected {self._job_name!r}, got {job_name!r}",
            )
        return None

    def _park(self, header: dict, item) -> None:
        key = (header["up"], header["down"])
        self.received_op_count += 1
        fut = self._slots.get(key)
        if fut is None or fut.done():
            fut = asyncio.get_running_loop().create_future()
            self._slots[key] = fut
        if not fut.done():
            fut.set_result(item)

    def deliver(self, kind: int, header: dict, payload: bytes) -> Tuple[int, str]:
        """Called by the transport on message arrival (on the I/O loop).
        Returns (code, result) for the ack."""
        bad = self.check_job(header)
        if bad is not None:
            return bad
        if (
            header.get("up") == constants.PING_SEQ_ID
            and header.get("down") == constants.PING_SEQ_ID
        ):
            # Readiness ping: ack without parking — nothing ever consumes
            # these, so parking would leak a slot per ping — and without
            # counting it as a received data op.
            return 200, "OK"
        self._park(header, (kind, header, payload))
        return 200, "OK"

    def deliver_obj(self, header: dict, obj) -> Tuple[int, str]:
        """Deliver an already-decoded object (the shm lane decodes before
        acking so the sender can recycle its segment on ack)."""
        bad = self.check_job(header)
        if bad is not None:
            return bad
        self._park(header, (self._KIND_OBJ, header, obj))
        return 200, "OK"

    def try_take(self, upstream_seq_id, curr_seq_id):
        """Opportunistic cross-thread take: return the parked item if its
        delivery already completed, else None.  Safe off-loop: dict ops are
        GIL-atomic, only the I/O loop creates/resolves slot futures, and a
        done future is immutable."""
        key = (str(upstream_seq_id), str(curr_seq_id))
        fut = self._slots.get(key)
        if fut is not None and fut.done() and not fut.cancelled():
            self._slots.pop(key, None)
            return fut.result()
        return None

    def consume_sync(self, item):
        """Deserialize a taken item in the CALLING thread (fast path that
        skips the I/O-loop round trip when data already arrived)."""
        kind, header, payload = item
        if kind == self._KIND_OBJ:
            if isinstance(payload, BaseException):
                raise payload
            return payload
        if kind == frames.KIND_ERROR:
            raise serialization.loads(payload, self._allowed_list)
        if kind == frames.KIND_TENSOR:
            return tensor_codec.decode(
                header, memoryview(payload), self.gpu_plane, self._allowed_list
            )
        return serialization.loads(payload, self._allowed_list)

    async def get_data(self, upstream_seq_id, curr_seq_id):
        key = (str(upstream_seq_id), str(curr_seq_id))
        fut = self._slots.get(key)
        if fut is None:
            fut = asyncio.get_running_loop().create_future()
            self._slots[key] = fut
        kind, header, payload = await fut
        self._slots.pop(key, None)
        loop = asyncio.get_running_loop()
        if kind == self._KIND_OBJ:
            if isinstance(payload, BaseException):
                raise payload
            return payload
        if kind == frames.KIND_ERROR:
            raise serialization.loads(payload, self._allowed_list)
        if kind == frames.KIND_TENSOR:
            if len(payload) >= OFFLOAD_BYTES:
                return await loop.run_in_executor(
                    None,
                    tensor_codec.decode,
                    header,
                    memoryview(payload),
                    self.gpu_plane,
                    self._allowed_list,
                )
            return tensor_codec.decode(
                header, memoryview(payload), self.gpu_plane, self._allowed_list
            )
        if len(payload) >= OFFLOAD_BYTES:
            return await loop.run_in_executor(
                None, serialization.loads, payload, self._allowed_list
            )
        return serialization.loads(payload, self._allowed_list)
