"""Default cross-silo transport: framed asyncio TCP with optional mutual TLS.

Why not gRPC for the hot path?  Python gRPC costs ≈1.3 ms per unary call on
loopback (measured here; sync and aio alike); a persistent framed asyncio-TCP
connection costs ≈0.1 ms cross-process.  The reference's per-iteration
latency is dominated by exactly this kind of overhead, so
the MI355X engine defaults to this transport and keeps the gRPC proxies
(``rayfed_amd.proxy.grpc``) as a pluggable, interop-friendly alternative via
``fed.init(sender_proxy_cls=..., receiver_proxy_cls=...)``.

Semantics preserved from the reference transport
(/root/reference/fed/proxy/grpc/grpc_proxy.py):

- same logical protocol (job-name multiplexed seq-id rendezvous, HTTP-style
  ack codes, 417 on job mismatch);
- same retry shape: the config's ``grpc_retry_policy`` (maxAttempts /
  initialBackoff / maxBackoff / backoffMultiplier) governs reconnect/retry,
  ``timeout_in_ms`` is the per-send deadline across attempts;
- TLS with optional mutual auth via the same ``tls_config`` dict
  (``ca_cert``/``key``/``cert`` (+ ``target_name_override``)).

Wire format (little-endian):
  request :  u64 length ‖ u64 request-id ‖ frame (frames.py)
  response:  u32 length ‖ msgpack {id, code, result}
The u64 request length admits >4 GiB tensor pushes; responses may arrive out
of order ("id" correlates them), so many sends pipeline on one connection.
Large payload parts (views over pinned GPU staging) are written sequentially
without any join copy.
"""
from __future__ import annotations

import asyncio
import itertools
import logging
import ssl
import time
from typing import Dict, Optional

import msgpack

from rayfed_amd import config as fed_config
from rayfed_amd.proxy import base_proxy
from rayfed_amd.proxy._encode import encode_request
from rayfed_amd.proxy._mailbox import Mailbox
from rayfed_amd.proxy.grpc import frames

logger = logging.getLogger(__name__)

_LEN = 4


def _parse_duration_s(v, default: float) -> float:
    if v is None:
        return default
    if isinstance(v, (int, float)):
        return float(v)
    s = str(v)
    if s.endswith("ms"):
        return float(s[:-2]) / 1000.0
    if s.endswith("s"):
        return float(s[:-1])
    return float(s)


class _RetryPolicy:
    def __init__(self, d: Optional[dict], proxy_max_restarts: Optional[int] = None):
        d = d or {}
        # proxy_max_restarts restarted the sender proxy ACTOR in the
        # reference (barriers.py:301-307); here the proxy is in-process, so
        # it maps to the reconnect budget per send: N restarts → N+1
        # attempts.  An explicit retry policy's maxAttempts wins.
        if "maxAttempts" in d or proxy_max_restarts is None:
            self.max_attempts = int(d.get("maxAttempts", 5))
        else:
            self.max_attempts = int(proxy_max_restarts) + 1
        self.initial_backoff = _parse_duration_s(d.get("initialBackoff"), 5.0)
        self.max_backoff = _parse_duration_s(d.get("maxBackoff"), 30.0)
        self.multiplier = float(d.get("backoffMultiplier", 2))


def _client_ssl_context(tls_config: dict) -> ssl.SSLContext:
    ctx = ssl.create_default_context(
        ssl.Purpose.SERVER_AUTH, cafile=tls_config.get("ca_cert")
    )
    if "cert" in tls_config and "key" in tls_config:
        ctx.load_cert_chain(tls_config["cert"], tls_config["key"])
    return ctx


def _server_ssl_context(tls_config: dict) -> ssl.SSLContext:
    ctx = ssl.create_default_context(ssl.Purpose.CLIENT_AUTH)
    ctx.load_cert_chain(tls_config["cert"], tls_config["key"])
    if tls_config.get("ca_cert"):
        ctx.load_verify_locations(cafile=tls_config["ca_cert"])
        ctx.verify_mode = ssl.CERT_REQUIRED  # mutual auth
    return ctx


class _Connection:
    """One persistent, pipelined connection to a destination party."""

    def __init__(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        self.reader = reader
        self.writer = writer
        self.pending: Dict[int, asyncio.Future] = {}
        self.ids = itertools.count(1)
        self.alive = True
        # Held across ALL writes of one frame: the chunk loop below awaits
        # drain() mid-frame, and without the lock a concurrent send coroutine
        # pipelining on this connection could interleave its bytes inside the
        # first request's frame, corrupting the length-prefixed stream.
        self.write_lock = asyncio.Lock()
        self.reader_task = asyncio.get_running_loop().create_task(self._read_loop())

    async def _read_loop(self):
        try:
            while True:
                hdr = await self.reader.readexactly(_LEN)
                n = int.from_bytes(hdr, "little")
                body = await self.reader.readexactly(n)
                resp = msgpack.unpackb(body, raw=False)
                fut = self.pending.pop(resp.get("id", 0), None)
                if fut is not None and not fut.done():
                    fut.set_result(resp)
        except (asyncio.IncompleteReadError, ConnectionError, OSError) as e:
            self._fail_all(e)
        except asyncio.CancelledError:
            self._fail_all(ConnectionError("connection closed"))

    def _fail_all(self, exc: Exception):
        self.alive = False
        for fut in self.pending.values():
            if not fut.done():
                fut.set_exception(ConnectionError(str(exc)))
        self.pending.clear()

    async def request(self, prefix: bytes, parts, timeout: float,
                      chunk: int = 8 << 20) -> dict:
        req_id = next(self.ids)
        fut = asyncio.get_running_loop().create_future()
        self.pending[req_id] = fut
        total = len(prefix) + sum(len(p) for p in parts)
        head = (total + 8).to_bytes(8, "little") + req_id.to_bytes(8, "little")
        _CHUNK = chunk  # bound transport buffering for multi-GiB parts
        try:
            async with self.write_lock:  # one frame's bytes stay contiguous
                if total <= 65536:
                    # Small request: one write, one TCP segment.
                    self.writer.write(
                        head + prefix + b"".join(bytes(p) for p in parts)
                    )
                else:
                    self.writer.write(head)
                    self.writer.write(prefix)
                    for p in parts:
                        if len(p) <= _CHUNK:
                            self.writer.write(p)
                        else:
                            mv = memoryview(p)
                            for off in range(0, len(mv), _CHUNK):
                                self.writer.write(mv[off : off + _CHUNK])
                                await self.writer.drain()
                await self.writer.drain()
            return await asyncio.wait_for(fut, timeout=timeout)
        finally:
            # A timed-out / cancelled / failed request must not leave its
            # future parked in ``pending`` for the connection's lifetime
            # (the ack for a retried id would otherwise resolve a dead slot).
            self.pending.pop(req_id, None)

    async def close(self):
        self.alive = False
        self.reader_task.cancel()
        try:
            self.writer.close()
            await self.writer.wait_closed()
        except Exception:  # noqa: BLE001
            pass


class TcpSenderProxy(base_proxy.SenderProxy):
    def __init__(self, addresses, party, job_name, tls_config, proxy_config=None):
        if proxy_config is not None and not isinstance(
            proxy_config, fed_config.CrossSiloMessageConfig
        ):
            proxy_config = fed_config.GrpcCrossSiloMessageConfig.from_dict(proxy_config)
        super().__init__(addresses, party, job_name, tls_config, proxy_config)
        self._conns: Dict[str, _Connection] = {}
        self._conn_locks: Dict[str, asyncio.Lock] = {}
        self._same_host_cache: Dict[str, bool] = {}
        self.gpu_plane = None
        self.last_sent_bytes = 0
        self._retry = _RetryPolicy(
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
        # messages_max_size_in_bytes chunks this transport's writes (the
        # reference uses it as the gRPC message cap, grpc_options.py:28-29).
        cap = proxy_config.messages_max_size_in_bytes if proxy_config else None
        self._write_chunk = min(8 << 20, cap) if cap else 8 << 20
        self._ssl = _client_ssl_context(tls_config) if tls_config else None
        # Verify the certificate against the destination host unless the user
        # pins a name explicitly (same semantics as the reference's
        # grpc.ssl_target_name_override, grpc_proxy.py:131-136) — a hard
        # 'localhost' default would let any cert with a localhost SAN
        # authenticate as any party on any host.
        self._server_hostname = (
            tls_config.get("target_name_override") if tls_config else None
        )

    async def _ensure_conn(self, dest_party: str) -> _Connection:
        conn = self._conns.get(dest_party)
        if conn is not None and conn.alive:
            return conn
        lock = self._conn_locks.setdefault(dest_party, asyncio.Lock())
        async with lock:
            conn = self._conns.get(dest_party)
            if conn is not None and conn.alive:
                return conn
            if dest_party not in self._addresses:
                raise ValueError(f"unknown dest party {dest_party!r}")
            host, port = self._addresses[dest_party].rsplit(":", 1)
            reader, writer = await asyncio.open_connection(
                host,
                int(port),
                ssl=self._ssl,
                server_hostname=(self._server_hostname or host) if self._ssl else None,
                limit=16 << 20,
            )
            writer.transport.set_write_buffer_limits(high=1 << 26)
            conn = _Connection(reader, writer)
            self._conns[dest_party] = conn
            return conn

    def _same_host(self, dest_party: str) -> bool:
        """The shm lane applies when the peer shares this node (loopback or
        local hostname) — the BASELINE topologies co-locate all parties."""
        cached = self._same_host_cache.get(dest_party)
        if cached is not None:
            return cached
        import socket

        host = self._addresses.get(dest_party, "").rsplit(":", 1)[0]
        same = host in ("127.0.0.1", "localhost", "::1", socket.gethostname())
        self._same_host_cache[dest_party] = same
        return same

    async def send(self, dest_party, data, upstream_seq_id, downstream_seq_id):
        from rayfed_amd.ops import shm_pool

        use_shm = shm_pool.shm_enabled() and self._same_host(dest_party)
        req = await encode_request(
            self._job_name, data, upstream_seq_id, downstream_seq_id,
            self.gpu_plane, shm=use_shm,
        )
        self.last_sent_bytes = req.total_len
        try:
            return await self._send_framed(dest_party, req)
        finally:
            req.release()  # pinned staging back to the pool after the ack

    async def _send_framed(self, dest_party, req):
        deadline = time.monotonic() + self._timeout_s
        backoff = self._retry.initial_backoff
        attempt = 0
        while True:
            attempt += 1
            try:
                conn = await self._ensure_conn(dest_party)
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    raise TimeoutError(f"send to {dest_party} deadline exceeded")
                resp = await conn.request(
                    req.prefix, req.parts, timeout=remaining,
                    chunk=self._write_chunk,
                )
                code = resp.get("code", 500)
                if 400 <= code < 500:
                    raise RuntimeError(
                        f"[{code}] send to {dest_party} rejected: "
                        f"{resp.get('result', '')}"
                    )
                if code >= 500:
                    raise RuntimeError(
                        f"[{code}] send to {dest_party} failed: "
                        f"{resp.get('result', '')}"
                    )
                return True
            except (ConnectionError, OSError, asyncio.TimeoutError, TimeoutError) as e:
                # UNAVAILABLE-equivalent: retry with backoff inside the deadline.
                now = time.monotonic()
                if attempt >= self._retry.max_attempts or now + backoff >= deadline:
                    raise RuntimeError(
                        f"send to {dest_party} failed after {attempt} attempts: {e!r}"
                    ) from e
                logger.debug(
                    "send to %s attempt %d failed (%r); retrying in %.1fs",
                    dest_party,
                    attempt,
                    e,
                    backoff,
                )
                await asyncio.sleep(backoff)
                backoff = min(backoff * self._retry.multiplier, self._retry.max_backoff)

    async def stop(self):
        for conn in self._conns.values():
            await conn.close()
        self._conns.clear()


class TcpReceiverProxy(base_proxy.ReceiverProxy):
    def __init__(self, listening_address, party, job_name, tls_config, proxy_config=None):
        if proxy_config is not None and not isinstance(
            proxy_config, fed_config.CrossSiloMessageConfig
        ):
            proxy_config = fed_config.GrpcCrossSiloMessageConfig.from_dict(proxy_config)
        super().__init__(listening_address, party, job_name, tls_config, proxy_config)
        self._server: Optional[asyncio.AbstractServer] = None
        self._mailbox = Mailbox(
            job_name,
            proxy_config.serializing_allowed_list if proxy_config else None,
        )

    @property
    def gpu_plane(self):
        return self._mailbox.gpu_plane

    @gpu_plane.setter
    def gpu_plane(self, plane):
        self._mailbox.gpu_plane = plane

    @property
    def received_op_count(self) -> int:
        return self._mailbox.received_op_count

    async def start(self):
        port = int(self._listening_address.rsplit(":", 1)[1])
        ssl_ctx = _server_ssl_context(self._tls_config) if self._tls_config else None
        try:
            self._server = await asyncio.start_server(
                self._handle_conn, host=None, port=port, ssl=ssl_ctx,
                reuse_address=False, limit=16 << 20,
            )
        except OSError as e:
            raise AssertionError(
                f"Failed to listen on port {port}: it is in use ({e}). "
                f"Choose another port in the cluster addresses."
            ) from e
        logger.info(
            "Receiver proxy of %s listening on %s (tls=%s)",
            self._party,
            port,
            ssl_ctx is not None,
        )

    async def _handle_conn(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        try:
            while True:
                hdr = await reader.readexactly(8)
                n = int.from_bytes(hdr, "little")
                body = await reader.readexactly(n)
                req_id = int.from_bytes(body[:8], "little")
                try:
                    # Zero-copy: the mailbox holds a view into the immutable
                    # request body until the reader consumes it.
                    kind, header, payload = frames.decode_frame(body[8:])
                    if kind == frames.KIND_TENSOR and any(
                        "shm" in m or "ipc_slabs" in m or "ipcp" in m or m.get("ipcg")
                        for m in header.get("tensors", ())
                    ):
                        # shm lane: consume (H2D + CRC) BEFORE acking — the
                        # ack licenses the sender to recycle its segment.
                        code, result = await self._consume_shm_frame(
                            header, payload
                        )
                    else:
                        code, result = self._mailbox.deliver(kind, header, payload)
                except ValueError as e:
                    code, result = 400, f"bad frame: {e}"
                if code == 417:
                    logger.warning("Rejected message: %s", result)
                resp = msgpack.packb(
                    {"id": req_id, "code": code, "result": result}, use_bin_type=True
                )
                writer.write(len(resp).to_bytes(_LEN, "little") + resp)
                await writer.drain()
        except (asyncio.IncompleteReadError, ConnectionError, OSError):
            pass
        except ssl.SSLError as e:  # plaintext client against TLS server, etc.
            logger.debug("TLS handshake/read failed: %r", e)
        finally:
            try:
                writer.close()
            except Exception:  # noqa: BLE001
                pass

    async def _consume_shm_frame(self, header, payload):
        bad = self._mailbox.check_job(header)
        if bad is not None:
            return bad
        from rayfed_amd.ops import tensor_codec

        loop = asyncio.get_running_loop()
        try:
            obj = await loop.run_in_executor(
                None,
                tensor_codec.decode,
                {k: header[k] for k in ("skel", "tensors", "ipc_group")
                 if k in header},
                memoryview(payload),
                self.gpu_plane,
                self._mailbox._allowed_list,
            )
        except Exception as e:  # noqa: BLE001 — CRC mismatch, attach failure…
            logger.warning("shm frame consume failed: %r", e)
            return 500, f"shm consume failed: {e!r}"
        return self._mailbox.deliver_obj(header, obj)

    async def get_data(self, src_party, upstream_seq_id, curr_seq_id):
        return await self._mailbox.get_data(upstream_seq_id, curr_seq_id)

    async def stop(self):
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None
