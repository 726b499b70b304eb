"""Python services over the C++ transport core (csrc/xfer_core.cpp).

The native tier of the cross-silo hot path: framing, sockets, routing, the
receive mailbox and ack round trips run in C++ threads with the GIL
released; Python handles (de)serialization and the shm-lane consume.
Selected automatically by ``rayfed_amd.proxy.barriers`` — plaintext AND TLS
(OpenSSL mutual auth in csrc/xfer_core.cpp) — whenever the extension is
built; ``RAYFED_TRANSPORT=asyncio`` forces the Python transport instead.

These classes implement the same *service* interface as
``barriers.SenderProxyService`` / ``ReceiverProxyService`` (send/get_data
returning concurrent futures, ``_get_stats``, ``stop``) — the coroutine SPI
does not apply because nothing here runs on an event loop.
"""
from __future__ import annotations

import logging
import threading
import time
from concurrent.futures import Future, ThreadPoolExecutor
from concurrent.futures import TimeoutError as FutureTimeoutError
from typing import Dict, Optional

from rayfed_amd import config as fed_config
from rayfed_amd.ops import tensor_codec
from rayfed_amd.proxy.grpc import frames
from rayfed_amd.runtime.object_ref import ObjectRef

logger = logging.getLogger(__name__)

_OBJ_MARKER = b"\x00RFOBJ"


def _load_xfer():
    import torch  # noqa: F401 — libc10 first

    import rayfed_amd._xfer as xfer

    return xfer


def xfer_available() -> bool:
    try:
        _load_xfer()
        return True
    except ImportError:
        return False


def _prestart_pool(pool: ThreadPoolExecutor, n: int) -> None:
    """Spawn all worker threads now: thread creation (~0.1 ms each) must not
    land inside the first benchmark rounds after a short warmup."""
    import threading

    barrier = threading.Barrier(n + 1)

    def _wait():
        try:
            barrier.wait(timeout=5)
        except threading.BrokenBarrierError:
            pass

    for _ in range(n):
        pool.submit(_wait)
    try:
        barrier.wait(timeout=5)
    except threading.BrokenBarrierError:
        pass


def _chunk_stripes() -> int:
    """Stripe fanout per sidecar chunk frame (parallel connections the
    chunk-streamed socket lane rides); RAYFED_CHUNK_STRIPES overrides."""
    import os

    try:
        return max(1, min(16, int(os.environ.get("RAYFED_CHUNK_STRIPES", "8"))))
    except ValueError:
        return 8


def _xfer_debug() -> bool:
    import os

    return os.environ.get("RAYFED_XFER_DEBUG") == "1"


def _coerce_config(proxy_config):
    if proxy_config is not None and not isinstance(
        proxy_config, fed_config.CrossSiloMessageConfig
    ):
        proxy_config = fed_config.GrpcCrossSiloMessageConfig.from_dict(proxy_config)
    return proxy_config


class _Retry:
    def __init__(self, d: Optional[dict], proxy_max_restarts: Optional[int] = None):
        from rayfed_amd.proxy.tcp.tcp_proxy import _RetryPolicy

        self._p = _RetryPolicy(d, proxy_max_restarts=proxy_max_restarts)

    @property
    def max_attempts(self):
        return self._p.max_attempts

    @property
    def initial_backoff(self):
        return self._p.initial_backoff

    @property
    def max_backoff(self):
        return self._p.max_backoff

    @property
    def multiplier(self):
        return self._p.multiplier


class XferSenderService:
    """send() → concurrent Future; encode + C++ socket I/O on pool threads."""

    def __init__(self, addresses: Dict, party: str, job_name: str,
                 proxy_config=None, tls_config=None):
        proxy_config = _coerce_config(proxy_config)
        self._addresses = addresses
        self._party = party
        self._job_name = job_name
        self._proxy_config = proxy_config
        xfer = _load_xfer()
        tls_kw = {}
        if tls_config:
            tls_kw = {
                "tls_ca": tls_config.get("ca_cert", ""),
                "tls_cert": tls_config.get("cert", ""),
                "tls_key": tls_config.get("key", ""),
                # Empty → the C++ core verifies against each destination's
                # host; set only to pin a name explicitly.
                "server_name": tls_config.get("target_name_override", ""),
            }
        # Two connections per destination: control frames must not queue
        # behind a multi-GiB defer-ack consume on the bulk lane (a float
        # broadcast measured 100 ms stuck behind a 16 GB frame's consume).
        self._tls = bool(tls_config)
        self._client_ctl = xfer.XferClient(job_name, **tls_kw)
        self._client_bulk = xfer.XferClient(job_name, **tls_kw)
        workers = (
            min(proxy_config.max_concurrency, 64)
            if proxy_config and proxy_config.max_concurrency
            else 8
        )
        self._pool = ThreadPoolExecutor(
            max_workers=workers, thread_name_prefix="xfer-send"
        )
        _prestart_pool(self._pool, workers)
        # Sidecar chunk frames of chunk-streamed sends (see _send_chunked).
        self._chunk_pool = ThreadPoolExecutor(
            max_workers=8, thread_name_prefix="xfer-chunk"
        )
        self._retry = _Retry(
            getattr(proxy_config, "grpc_retry_policy", None) if proxy_config else None,
            proxy_max_restarts=(
                proxy_config.proxy_max_restarts if proxy_config else None
            ),
        )
        self._timeout_s = (
            (proxy_config.timeout_in_ms / 1000.0)
            if proxy_config and proxy_config.timeout_in_ms
            else 60.0
        )
        # Frames above one stripe's worth split across parallel connections
        # (cross-host lane: one TCP stream tops out ~2 GB/s).  The stripe
        # size honors messages_max_size_in_bytes when set — the message-cap
        # knob chunks on this transport like it caps gRPC messages.
        self._stripe_bytes = (
            proxy_config.messages_max_size_in_bytes
            if proxy_config and proxy_config.messages_max_size_in_bytes
            else (256 << 20)
        )
        self.gpu_plane = None
        self._stats_lock = threading.Lock()
        self.send_op_count = 0
        self._same_host_cache: Dict[str, bool] = {}
        from rayfed_amd.proxy.barriers import _EdgeStats

        self._edges: Dict[str, object] = {}
        self._edge_cls = _EdgeStats

    # service interface -------------------------------------------------------
    @property
    def proxy(self):
        return self  # config introspection parity (tests read ._proxy_config)

    def _same_host(self, dest_party: str) -> bool:
        cached = self._same_host_cache.get(dest_party)
        if cached is not None:
            return cached
        import socket

        host = self._addresses.get(dest_party, "").rsplit(":", 1)[0]
        same = host in ("127.0.0.1", "localhost", "::1", socket.gethostname())
        self._same_host_cache[dest_party] = same
        return same

    # Frames at or below this ride the inline fast path: written by the
    # thread that produced the data (send_async), acked lazily.
    _INLINE_MAX = 256 << 10

    def send(self, dest_party, data, upstream_seq_id, downstream_seq_id) -> Future:
        with self._stats_lock:
            self.send_op_count += 1
        up = str(upstream_seq_id)
        down = str(downstream_seq_id)
        if self._tls:
            return self._send_tls_continuation(dest_party, data, up, down)

        # Continuation-style send: issue the frame the moment the payload is
        # ready, IN the producing thread (already hot) — a pool thread woken
        # from idle costs 60-150 us, measured dominant on the tiny-task path.
        state: dict = {}
        ready = threading.Event()

        def _issue(_f=None):
            try:
                d = data
                if isinstance(d, ObjectRef):
                    d = d.result()  # done (callback path); error propagates
                body_parts, extras, defer_ack, nbytes = self._encode_frame(
                    dest_party, d, up, down
                )
                if not defer_ack and nbytes <= self._INLINE_MAX:
                    from rayfed_amd._private import tracing

                    if tracing.enabled:
                        tracing.event("xfer.send_inline", "xsilo",
                                      dest=dest_party, up=up, down=down)
                    host, port = self._addresses[dest_party].rsplit(":", 1)
                    state["t0"] = time.perf_counter()
                    state.update(
                        parts=body_parts, extras=extras, nbytes=nbytes,
                        host=host, port=int(port),
                    )
                    try:
                        state["handle"] = self._client_ctl.send_async(
                            host, int(port), up, down, body_parts, False
                        )
                    except RuntimeError:
                        # e.g. peer not up yet: the pooled path carries the
                        # retry/backoff budget (async-startup semantics).
                        state.pop("handle", None)
                        state["fut"] = self._pool.submit(
                            self._send_parts, dest_party, body_parts,
                            extras, False, nbytes, up, down,
                        )
                else:
                    # Big / DEFER_ACK frame: ship the ALREADY-ENCODED parts
                    # on the pool — re-encoding would pack (and leak) a
                    # second set of staging slabs.
                    state["fut"] = self._pool.submit(
                        self._send_parts, dest_party, body_parts, extras,
                        defer_ack, nbytes, up, down,
                    )
            except BaseException as e:  # noqa: BLE001
                state["exc"] = e
            finally:
                ready.set()

        def _fetch(timeout=None):
            if not ready.wait(timeout):
                raise FutureTimeoutError()
            if "exc" in state:
                raise state["exc"]
            if "fut" in state:
                return state["fut"].result(timeout)
            err = True
            try:
                try:
                    code, result = self._client_ctl.wait_ack(
                        state["handle"], self._timeout_s
                    )
                except RuntimeError:
                    # Connection broke / ack lost: blocking re-send with the
                    # retry budget (frames are idempotent — the mailbox is
                    # keyed by seq ids).
                    code, result = self._send_with_retry(
                        self._client_ctl, state["host"], state["port"],
                        up, down, state["parts"], False,
                    )
                if 400 <= code < 500:
                    raise RuntimeError(
                        f"[{code}] send to {dest_party} rejected: {result}"
                    )
                if code >= 500:
                    raise RuntimeError(
                        f"[{code}] send to {dest_party} failed: {result}"
                    )
                err = False
                return True
            finally:
                tensor_codec.release_parts(state["extras"])
                secs = time.perf_counter() - state["t0"]
                with self._stats_lock:
                    edge = self._edges.setdefault(dest_party, self._edge_cls())
                    edge.record(state["nbytes"], secs, err)

        if isinstance(data, ObjectRef) and not data.future.done():
            data.future.add_done_callback(_issue)
        else:
            _issue()
        return _LazyFuture(_fetch)

    def _send_tls_continuation(self, dest_party, data, up, down) -> Future:
        """TLS lane: no pipelined send_async (an SSL* is exclusive), so a
        small frame's whole write+ack exchange runs inline in the producing
        thread — one quick attempt, pool fallback with the full retry
        budget.  Saves the pool wake-from-idle like the plaintext path."""
        state: dict = {}
        ready = threading.Event()

        def _issue(_f=None):
            try:
                d = data
                if isinstance(d, ObjectRef):
                    d = d.result()
                body_parts, extras, defer_ack, nbytes = self._encode_frame(
                    dest_party, d, up, down
                )
                if not defer_ack and nbytes <= self._INLINE_MAX:
                    t0 = time.perf_counter()
                    host, port = self._addresses[dest_party].rsplit(":", 1)
                    try:
                        code, result = self._client_ctl.send(
                            host, int(port), up, down, body_parts, False, 2.0
                        )
                        tensor_codec.release_parts(extras)
                        state["inline"] = (code, result, nbytes, t0)
                        return
                    except RuntimeError:
                        pass  # cold/broken conn: pooled path retries
                state["fut"] = self._pool.submit(
                    self._send_parts, dest_party, body_parts, extras,
                    defer_ack, nbytes, up, down,
                )
            except BaseException as e:  # noqa: BLE001
                state["exc"] = e
            finally:
                ready.set()

        def _fetch(timeout=None):
            if not ready.wait(timeout):
                raise FutureTimeoutError()
            if "exc" in state:
                raise state["exc"]
            if "fut" in state:
                return state["fut"].result(timeout)
            code, result, nbytes, t0 = state["inline"]
            err = not (200 <= code < 400)
            with self._stats_lock:
                edge = self._edges.setdefault(dest_party, self._edge_cls())
                edge.record(nbytes, time.perf_counter() - t0, err)
            if 400 <= code < 500:
                raise RuntimeError(
                    f"[{code}] send to {dest_party} rejected: {result}"
                )
            if code >= 500:
                raise RuntimeError(
                    f"[{code}] send to {dest_party} failed: {result}"
                )
            return True

        if isinstance(data, ObjectRef) and not data.future.done():
            data.future.add_done_callback(_issue)
        else:
            _issue()
        return _LazyFuture(_fetch)

    def _encode_frame(self, dest_party, data, up, down):
        """Serialize one payload into wire parts.  Returns
        (body_parts, extras, defer_ack, nbytes)."""
        from rayfed_amd.exceptions import FedRemoteError
        from rayfed_amd.ops import shm_pool

        header = {"job": self._job_name, "up": up, "down": down}
        if isinstance(data, FedRemoteError):
            from rayfed_amd._private import serialization

            extras = {"tensors": []}
            body_parts = [
                frames.encode_frame_prefix(frames.KIND_ERROR, header),
                serialization.dumps(data),
            ]
            defer_ack = False
        else:
            use_shm = shm_pool.shm_enabled() and self._same_host(dest_party)
            extras, parts = tensor_codec.encode(data, self.gpu_plane, use_shm)
            defer_ack = False
            if extras["tensors"]:
                wire_header = {
                    k: v for k, v in extras.items() if k != "_releases"
                }
                header.update(wire_header)
                defer_ack = any(
                    "shm" in m or "ipc_slabs" in m or "ipcp" in m or m.get("ipcg")
                    for m in extras["tensors"]
                )
                body_parts = [
                    frames.encode_frame_prefix(frames.KIND_TENSOR, header)
                ] + list(parts)
            else:
                body_parts = [
                    frames.encode_frame_prefix(frames.KIND_PICKLE, header),
                    parts[0],
                ]
        nbytes = sum(len(p) for p in body_parts)
        return body_parts, extras, defer_ack, nbytes

    def _send_blocking(self, dest_party, data, up, down) -> bool:
        from rayfed_amd._private import tracing

        if tracing.enabled:
            with tracing.span("xfer.send", "xsilo", dest=dest_party, up=up,
                              down=down):
                return self._send_blocking_inner(dest_party, data, up, down)
        return self._send_blocking_inner(dest_party, data, up, down)

    def _send_blocking_inner(self, dest_party, data, up, down) -> bool:
        if isinstance(data, ObjectRef):
            data = data.result()  # producer error propagates to the future
        try:
            body_parts, extras, defer_ack, nbytes = self._encode_frame(
                dest_party, data, up, down
            )
        except BaseException:
            with self._stats_lock:
                edge = self._edges.setdefault(dest_party, self._edge_cls())
                edge.record(0, 0.0, True)
            raise
        return self._send_parts(
            dest_party, body_parts, extras, defer_ack, nbytes, up, down
        )

    @staticmethod
    def _slice_parts(parts, lo, hi):
        """Byte range [lo, hi) across a list of buffer parts (zero-copy)."""
        out = []
        pos = 0
        for p in parts:
            mv = memoryview(p)
            plo, phi = pos, pos + len(mv)
            pos = phi
            if phi <= lo or plo >= hi:
                continue
            out.append(mv[max(plo, lo) - plo : min(phi, hi) - plo])
        return out

    def _send_parts(self, dest_party, body_parts, extras, defer_ack, nbytes,
                    up, down) -> bool:
        """Ship pre-encoded frame parts with retry; releases staging after
        the ack and records edge stats."""
        t0 = time.perf_counter()
        err = True
        try:
            try:
                host, port = self._addresses[dest_party].rsplit(":", 1)
                prefix = bytes(body_parts[0][:6]) if body_parts else b""
                payload_len = nbytes - (len(body_parts[0]) if body_parts else 0)
                if (
                    not defer_ack
                    and len(prefix) == 6
                    and prefix[:4] == frames.MAGIC
                    and prefix[5] == frames.KIND_TENSOR
                    and payload_len >= 2 * self._chunk_unit(payload_len)
                ):
                    # Applies to TLS too: the sidecar chunk frames stripe
                    # across parallel TLS connections (stripe_tls), so the
                    # crypto parallelizes with the pinned in-place assembly
                    # and the consumer-side H2D overlap — measured 1.8 ->
                    # multi-GB/s on the 2 GiB TLS push vs the whole-frame
                    # path this branch replaces.
                    # Chunk-streamed tensor frame: the receiver H2Ds each
                    # chunk as it lands, overlapping consume with arrival
                    # (a whole-frame send serializes network then consume).
                    if _xfer_debug():
                        import sys as _sys

                        print(f"[xfer-tx] chunked send {up}/{down} "
                              f"payload={payload_len>>20}MiB t={time.monotonic():.3f}",
                              file=_sys.stderr, flush=True)
                    code, result = self._send_chunked(
                        host, int(port), up, down, body_parts, payload_len
                    )
                    if _xfer_debug():
                        import sys as _sys

                        print(f"[xfer-tx] chunked done {up}/{down} "
                              f"t={time.monotonic():.3f}",
                              file=_sys.stderr, flush=True)
                else:
                    bulk = defer_ack or nbytes > (1 << 20)
                    stripes = 1
                    if not defer_ack and nbytes >= 2 * self._stripe_bytes:
                        stripes = min(8, -(-nbytes // self._stripe_bytes))
                    code, result = self._send_with_retry(
                        self._client_bulk if bulk else self._client_ctl,
                        host, int(port), up, down, body_parts, defer_ack,
                        stripes=stripes,
                    )
            finally:
                tensor_codec.release_parts(extras)
            if 400 <= code < 500:
                raise RuntimeError(
                    f"[{code}] send to {dest_party} rejected: {result}"
                )
            if code >= 500:
                raise RuntimeError(
                    f"[{code}] send to {dest_party} failed: {result}"
                )
            err = False
            return True
        finally:
            secs = time.perf_counter() - t0
            with self._stats_lock:
                edge = self._edges.setdefault(dest_party, self._edge_cls())
                edge.record(nbytes, secs, err)

    def _chunk_unit(self, payload_len: int) -> int:
        """Chunk size for the chunk-streamed path: an eighth of the payload
        (so mid-size frames parallelize too — a single TCP stream tops out
        ~2.35 GB/s), floored at 8 MiB (sub-frame overhead) and capped by
        the configured messages_max_size stripe."""
        return min(self._stripe_bytes, max(8 << 20, payload_len // 8))

    def _send_chunked(self, host, port, up, down, body_parts, payload_len):
        """Split a tensor frame's payload into sidecar chunk frames sent in
        parallel (each striped over the stripe connections), plus a small
        KIND_CHUNKED main frame carrying the inner frame's prefix."""
        C = self._chunk_unit(payload_len)
        K = min(32, -(-payload_len // C))
        C = -(-payload_len // K)  # rebalance so every chunk is ~equal
        payload_parts = list(body_parts[1:])
        futs = []
        for i in range(K):
            sub = self._slice_parts(payload_parts, i * C, min((i + 1) * C,
                                                              payload_len))
            # Dedicated pool: _send_parts itself runs on self._pool, and
            # waiting there for chunk futures queued to the same pool could
            # starve under concurrent big sends.
            futs.append(self._chunk_pool.submit(
                self._send_with_retry, self._client_bulk, host, port,
                f"{up}\x01x{i}", down, sub, False, _chunk_stripes(), True,
            ))
        meta = {
            "job": self._job_name, "up": up, "down": down,
            "xk": K, "xc": C, "xlen": payload_len,
        }
        main = [
            frames.encode_frame_prefix(frames.KIND_CHUNKED, meta),
            body_parts[0],
        ]
        try:
            code, result = self._send_with_retry(
                self._client_ctl, host, port, up, down, main, False
            )
        finally:
            # ALWAYS join the chunk senders — the caller releases the
            # staging these threads are still reading the moment we return
            # (even on the main-frame failure path).
            errs = []
            for f in futs:
                try:
                    c, r = f.result()
                except BaseException as e:  # noqa: BLE001
                    errs.append(e)
                    continue
                errs.append(None if c == 200 else RuntimeError(f"[{c}] {r}"))
        for e in errs:
            if e is not None:
                raise e
        return code, result

    def _send_with_retry(self, client, host, port, up, down, parts, defer_ack,
                         stripes: int = 1, pinned: bool = False):
        deadline = time.monotonic() + self._timeout_s
        backoff = self._retry.initial_backoff
        attempt = 0
        while True:
            attempt += 1
            try:
                remaining = max(0.001, deadline - time.monotonic())
                return client.send(
                    host, port, up, down, parts, defer_ack, remaining,
                    stripes, pinned,
                )
            except RuntimeError as e:
                now = time.monotonic()
                if attempt >= self._retry.max_attempts or now + backoff >= deadline:
                    raise RuntimeError(
                        f"send failed after {attempt} attempts: {e}"
                    ) from e
                logger.debug("xfer send attempt %d failed (%r)", attempt, e)
                time.sleep(backoff)
                backoff = min(backoff * self._retry.multiplier,
                              self._retry.max_backoff)

    def _get_stats(self) -> Dict[str, object]:
        with self._stats_lock:
            return {
                "send_op_count": self.send_op_count,
                "edges": {p: e.as_dict() for p, e in self._edges.items()},
            }

    def stop(self):
        self._pool.shutdown(wait=False, cancel_futures=True)
        self._chunk_pool.shutdown(wait=False, cancel_futures=True)
        self._client_ctl.close_all()
        self._client_bulk.close_all()


class XferReceiverService:
    """get_data() → concurrent Future; C++ mailbox, Python deserialization."""

    def __init__(self, listening_address: str, party: str, job_name: str,
                 proxy_config=None, tls_config=None):
        proxy_config = _coerce_config(proxy_config)
        self._party = party
        self._job_name = job_name
        self._proxy_config = proxy_config
        self._allowed_list = (
            proxy_config.serializing_allowed_list if proxy_config else None
        )
        port = int(listening_address.rsplit(":", 1)[1])
        xfer = _load_xfer()
        tls_kw = {}
        if tls_config:
            tls_kw = {
                "tls_cert": tls_config.get("cert", ""),
                "tls_key": tls_config.get("key", ""),
                "tls_ca": tls_config.get("ca_cert", ""),
            }
        try:
            self._server = xfer.XferServer(port, job_name, **tls_kw)
            self._server.start(self._consume_deferred)
        except RuntimeError as e:
            raise AssertionError(
                f"Failed to listen on port {port}: it is in use ({e})."
            ) from e
        workers = (
            min(proxy_config.max_concurrency, 256)
            if proxy_config and proxy_config.max_concurrency
            else 32
        )
        self._pool = ThreadPoolExecutor(
            max_workers=workers, thread_name_prefix="xfer-recv"
        )
        # Deferred consumes (H2D + CRC of shm/IPC frames) run here, off the
        # C++ connection threads, so a multi-GiB consume never stalls other
        # frames on the same connection.
        self._consume_pool = ThreadPoolExecutor(
            max_workers=4, thread_name_prefix="xfer-consume"
        )
        self._objs: Dict[tuple, object] = {}
        self._objs_lock = threading.Lock()
        self.gpu_plane = None
        self._stats_lock = threading.Lock()
        self.receive_op_count = 0
        self._deferred_count = 0

    @property
    def proxy(self):
        return self

    @property
    def received_op_count(self) -> int:
        return int(self._server.received_op_count) + self._deferred_count

    # C++ calls this (with the GIL) for DEFER_ACK frames.  It must return
    # immediately: the decode (H2D + CRC) runs on the consume pool and acks
    # via server.complete(token) when done — consume-before-ack preserved,
    # connection thread never blocked.
    def _consume_deferred(self, up: str, down: str, body: bytes, token: int):
        self._consume_pool.submit(self._consume_one, up, down, body, token)

    def _consume_one(self, up: str, down: str, body: bytes, token: int):
        try:
            kind, header, payload = frames.decode_frame(body)
            obj = tensor_codec.decode(
                {k: header[k] for k in ("skel", "tensors", "ipc_group")
                 if k in header},
                memoryview(payload),
                self.gpu_plane,
                self._allowed_list,
                allow_lazy=True,  # the ack can wait for lazy consumption
            )
        except Exception as e:  # noqa: BLE001
            logger.warning("xfer deferred consume failed: %r", e)
            self._server.complete(token, 500, f"consume failed: {e!r}")
            return
        lazies = (
            self.gpu_plane.pop_pending_lazies()
            if self.gpu_plane is not None
            else []
        )
        with self._objs_lock:
            self._objs[(up, down)] = obj
            self._deferred_count += 1
        self._server.post(up, down, _OBJ_MARKER)
        if not lazies:
            self._server.complete(token, 200)
            return
        # Zero-copy receive: the sender's slabs stay licensed to us until
        # every lazy handle is released; the LAST release sends the ack.
        remaining = [len(lazies)]
        lock = threading.Lock()

        def _one_done():
            with lock:
                remaining[0] -= 1
                done = remaining[0] == 0
            if done:
                self._server.complete(token, 200)

        for lz in lazies:
            lz._attach_completer(_one_done)

    def _take(self, up: str, down: str, body: bytes):
        if bytes(body) == _OBJ_MARKER:
            with self._objs_lock:
                return self._objs.pop((up, down))
        kind, header, payload = frames.decode_frame(body)
        from rayfed_amd._private import serialization

        if kind == frames.KIND_ERROR:
            raise serialization.loads(payload, self._allowed_list)
        if kind == frames.KIND_TENSOR:
            return tensor_codec.decode(
                header, memoryview(payload), self.gpu_plane, self._allowed_list
            )
        if kind == frames.KIND_CHUNKED:
            _ik, inner_header, _ = frames.decode_frame(payload)
            return tensor_codec.decode_streamed(
                inner_header, header["xlen"], header["xc"],
                self._iter_chunks(up, down, header["xk"]),
                self.gpu_plane, self._allowed_list,
            )
        return serialization.loads(payload, self._allowed_list)

    def _iter_chunks(self, up: str, down: str, k: int):
        """Yield sidecar chunk frames of a chunk-streamed send in ARRIVAL
        order (consume overlaps the network).  If the sender aborts, the
        substituted error object lands under the MAIN seq ids — poll for it
        so the consumer never hangs on a missing chunk."""
        import time as _time

        remaining = set(range(k))
        deadline = _time.monotonic() + 600
        while remaining:
            got = None
            for i in sorted(remaining):
                b = self._server.try_view(f"{up}\x01x{i}", down)
                if b is not None:
                    got = (i, b)
                    break
            if got is None:
                err = self._server.try_take(up, down)
                if err is not None:
                    self._take(up, down, err)  # raises for error frames
                    raise RuntimeError(
                        "unexpected non-error frame during chunked receive"
                    )
                if _time.monotonic() > deadline:
                    raise TimeoutError("chunk-streamed receive stalled")
                i = min(remaining)
                try:
                    b = self._server.wait_view(f"{up}\x01x{i}", down, 0.25)
                    got = (i, b)
                except RuntimeError as e:
                    if "timeout" in str(e):
                        continue
                    raise
            remaining.discard(got[0])
            if _xfer_debug():
                import sys as _sys

                print(f"[xfer-rx] chunk {got[0]} ({len(got[1])>>20}MiB) "
                      f"t={_time.monotonic():.3f}",
                      file=_sys.stderr, flush=True)
            yield got

    def get_data(self, src_party, upstream_seq_id, curr_seq_id) -> Future:
        up, down = str(upstream_seq_id), str(curr_seq_id)
        with self._stats_lock:
            self.receive_op_count += 1
        body = self._server.try_take(up, down)
        if body is not None:
            fut: Future = Future()
            try:
                fut.set_result(self._take(up, down, body))
            except BaseException as e:  # noqa: BLE001
                fut.set_exception(e)
            return fut

        def _wait(timeout=None):
            # Parity with the reference's recv semantics: wait indefinitely
            # for the peer's push (failure paths deliver an error object on
            # the same seq ids instead of leaving this hanging).  Slice the
            # C++ wait so server stop still unblocks promptly.
            deadline = None if timeout is None else time.monotonic() + timeout
            while True:
                if deadline is None:
                    chunk = 60.0
                else:
                    chunk = min(60.0, deadline - time.monotonic())
                    if chunk <= 0:
                        raise FutureTimeoutError()
                try:
                    b = self._server.get_data(up, down, chunk)
                    break
                except RuntimeError as e:
                    if "timeout" in str(e):
                        continue
                    raise  # server stopped
            return self._take(up, down, b)

        # Lazy: the consumer's own thread performs the C++ mailbox wait on
        # result() — no recv-pool hop (one futex wake-from-idle fewer per
        # cross-party edge on the tiny-task critical path).
        return _LazyFuture(_wait)

    def _get_stats(self) -> Dict[str, int]:
        return {"receive_op_count": self.receive_op_count}

    def stop(self):
        self._server.stop()
        self._pool.shutdown(wait=False, cancel_futures=True)
        self._consume_pool.shutdown(wait=False, cancel_futures=True)


class _LazyFuture(Future):
    """A Future whose value is produced by the first result() caller running
    ``fetch(timeout)`` in its own thread; other waiters block on the Future
    proper.  ``fetch`` raising concurrent.futures.TimeoutError leaves the
    future unresolved (standard result(timeout) semantics)."""

    def __init__(self, fetch):
        super().__init__()
        self._fetch = fetch
        self._claim = threading.Lock()

    def result(self, timeout=None):
        if not self.done() and self._claim.acquire(blocking=False):
            try:
                if not self.done():
                    try:
                        value = self._fetch(timeout)
                    except FutureTimeoutError:
                        raise  # not an outcome: the caller just gave up
                    except BaseException as e:  # noqa: BLE001
                        self.set_exception(e)
                    else:
                        self.set_result(value)
            finally:
                self._claim.release()
        return super().result(timeout)

    def exception(self, timeout=None):
        if not self.done():
            try:
                self.result(timeout)
            except BaseException:  # noqa: BLE001
                pass
        return super().exception(timeout)
