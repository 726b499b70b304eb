"""Round-2 hardening regressions: ping slot hygiene, honored config knobs,
concurrent big-frame safety on both transports.

Coverage targets: reference barriers.py:497-523 (ping), barriers.py:301-307
(proxy_max_restarts semantics), plus this engine's own pipelined-connection
invariants that the reference (one request per gRPC call) cannot violate.
"""
import threading

import pytest

import rayfed_amd.proxy.barriers as barriers
from rayfed_amd._private import constants
from rayfed_amd._private.global_context import (
    clear_global_context,
    init_global_context,
)
from rayfed_amd.config import GrpcCrossSiloMessageConfig
from tests._util import make_addresses


@pytest.fixture(params=["tcp", "xfer"])
def loop_env(request, monkeypatch):
    """Sender+receiver services for one party on loopback, per transport."""
    if request.param == "xfer":
        from rayfed_amd.proxy.xfer import xfer_available

        if not xfer_available():
            pytest.skip("C++ transport extension not built")
        monkeypatch.setenv("RAYFED_TRANSPORT", "cpp")
        sender_cls = receiver_cls = None  # barriers auto-selects the C++ core
    else:
        monkeypatch.setenv("RAYFED_TRANSPORT", "asyncio")
        sender_cls = receiver_cls = None
    addrs = make_addresses(["alice"])
    init_global_context(current_party="alice", job_name="r2_job")
    receiver = barriers.start_receiver_proxy(
        addrs, "alice", job_name="r2_job", proxy_cls=receiver_cls,
        proxy_config=None,
    )
    sender = barriers.start_sender_proxy(
        addrs, "alice", job_name="r2_job", proxy_cls=sender_cls,
        proxy_config=None,
    )
    yield request.param, addrs, sender, receiver
    clear_global_context()
    barriers._cleanup_proxies()


def test_ping_does_not_leak_mailbox_slots(loop_env):
    """Repeated pings must be acked without parking payloads or counting as
    received data ops (round-1 leak: every ping parked an unconsumed slot)."""
    kind, addrs, sender, receiver = loop_env
    for _ in range(5):
        ok = sender.send(
            "alice", b"data", constants.PING_SEQ_ID, constants.PING_SEQ_ID
        ).result(timeout=10)
        assert ok is True
    assert receiver.proxy.received_op_count == 0
    # Nothing parked under the ping ids on either mailbox flavor.
    if kind == "xfer":
        assert receiver._server.try_take(
            constants.PING_SEQ_ID, constants.PING_SEQ_ID
        ) is None
    else:
        assert receiver.proxy._mailbox.try_take(
            constants.PING_SEQ_ID, constants.PING_SEQ_ID
        ) is None
    # Real data still flows after pings.
    sender.send("alice", 41, "7", "7").result(timeout=10)
    assert receiver.get_data("alice", "7", "7").result(timeout=10) == 41


def test_ping_others_end_to_end(loop_env):
    kind, addrs, sender, receiver = loop_env
    assert barriers.ping_others(addrs, self_party="__nobody__") is True
    assert receiver.proxy.received_op_count == 0


def test_concurrent_large_sends_one_connection(loop_env):
    """Two >8 MiB frames pipelined to one destination must each arrive
    intact (regression: the asyncio transport could interleave chunked
    writes of concurrent frames mid-stream before the write lock)."""
    import numpy as np

    kind, addrs, sender, receiver = loop_env
    a = np.arange(3 << 20, dtype=np.uint8)          # 3 MiB
    b = (np.arange(12 << 20, dtype=np.uint8) * 7)   # 12 MiB > chunk threshold
    errs = []

    def _send(tag, arr):
        try:
            assert sender.send("alice", arr, tag, tag).result(timeout=30)
        except BaseException as e:  # noqa: BLE001
            errs.append(e)

    threads = [
        threading.Thread(target=_send, args=("100", a)),
        threading.Thread(target=_send, args=("101", b)),
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errs, errs
    out_a = receiver.get_data("alice", "100", "100").result(timeout=30)
    out_b = receiver.get_data("alice", "101", "101").result(timeout=30)
    assert np.array_equal(out_a, a)
    assert np.array_equal(out_b, b)


def test_concurrent_shm_deferred_consumes(loop_env):
    """Two shm-lane (DEFER_ACK) frames in flight at once: with the consume
    moved off the C++ connection thread both must complete, and a control
    frame sent mid-consume must not be blocked behind them."""
    torch = pytest.importorskip("torch")
    kind, addrs, sender, receiver = loop_env
    t1 = torch.arange(2 << 20, dtype=torch.float32)  # 8 MiB — rides shm
    t2 = torch.arange(2 << 20, dtype=torch.float32) * 2
    errs = []

    def _send(tag, t):
        try:
            assert sender.send("alice", t, tag, tag).result(timeout=30)
        except BaseException as e:  # noqa: BLE001
            errs.append(e)

    threads = [
        threading.Thread(target=_send, args=("201", t1)),
        threading.Thread(target=_send, args=("202", t2)),
        threading.Thread(target=_send, args=("203", {"ctl": 1})),
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errs, errs
    assert receiver.get_data("alice", "203", "203").result(timeout=30) == {
        "ctl": 1
    }
    assert torch.equal(receiver.get_data("alice", "201", "201").result(30), t1)
    assert torch.equal(receiver.get_data("alice", "202", "202").result(30), t2)
    from rayfed_amd.ops import shm_pool

    shm_pool.detach_all()
    # Drain the sender pool too — later tests assert exact segment reuse and
    # must not inherit this test's 8 MiB segments.
    shm_pool.get_send_pool().shutdown()


# ---------------------------------------------------------------- knob tests
def test_proxy_max_restarts_sets_reconnect_budget():
    from rayfed_amd.proxy.tcp.tcp_proxy import TcpSenderProxy

    cfg = GrpcCrossSiloMessageConfig.from_dict({"proxy_max_restarts": 2})
    p = TcpSenderProxy({"alice": "127.0.0.1:1"}, "alice", "j", None, cfg)
    assert p._retry.max_attempts == 3  # N restarts -> N+1 attempts

    # An explicit retry policy's maxAttempts wins over proxy_max_restarts.
    cfg2 = GrpcCrossSiloMessageConfig.from_dict(
        {"proxy_max_restarts": 2, "grpc_retry_policy": {"maxAttempts": 7}}
    )
    p2 = TcpSenderProxy({"alice": "127.0.0.1:1"}, "alice", "j", None, cfg2)
    assert p2._retry.max_attempts == 7


def test_max_concurrency_bounds_xfer_pools():
    from rayfed_amd.proxy.xfer import xfer_available

    if not xfer_available():
        pytest.skip("C++ transport extension not built")
    from rayfed_amd.proxy.xfer import XferReceiverService, XferSenderService

    addrs = make_addresses(["alice"])
    cfg = GrpcCrossSiloMessageConfig.from_dict({"max_concurrency": 3})
    recv = XferReceiverService(addrs["alice"], "alice", "j", cfg)
    send = XferSenderService(addrs, "alice", "j", cfg)
    try:
        assert recv._pool._max_workers == 3
        assert send._pool._max_workers == 3
    finally:
        send.stop()
        recv.stop()


def test_resource_labels_warn():
    import logging

    import rayfed_amd as fed

    records = []

    class _Capture(logging.Handler):
        def emit(self, record):
            records.append(record.getMessage())

    api_logger = logging.getLogger("rayfed_amd.api")
    handler = _Capture(level=logging.WARNING)
    api_logger.addHandler(handler)
    addrs = make_addresses(["alice"])
    try:
        fed.init(
            addresses=addrs,
            party="alice",
            config={
                "cross_silo_comm": {
                    "send_resource_label": {"node": "a"},
                    "recv_resource_label": {"node": "b"},
                }
            },
        )
        text = "\n".join(records)
        assert "send_resource_label" in text
        assert "recv_resource_label" in text
        assert "ignored" in text
    finally:
        api_logger.removeHandler(handler)
        fed.shutdown()


def test_striped_send_roundtrip():
    """Frames above 2x messages_max_size_in_bytes split across parallel
    connections (cross-host lane); content must reassemble exactly."""
    import numpy as np

    from rayfed_amd.proxy.xfer import xfer_available

    if not xfer_available():
        pytest.skip("C++ transport extension not built")
    from rayfed_amd.proxy.xfer import XferReceiverService, XferSenderService

    addrs = make_addresses(["alice"])
    cfg = GrpcCrossSiloMessageConfig.from_dict(
        {"messages_max_size_in_bytes": 1 << 20}
    )
    recv = XferReceiverService(addrs["alice"], "alice", "j", cfg)
    send = XferSenderService(addrs, "alice", "j", cfg)
    try:
        rng = np.random.default_rng(7)
        payload = rng.integers(0, 256, size=7 * (1 << 20) + 333, dtype=np.uint8)
        assert send.send("alice", payload, "400", "400").result(timeout=60)
        out = recv.get_data("alice", "400", "400").result(timeout=60)
        assert np.array_equal(out, payload)
        # A small frame after the striped one still flows on the same lanes.
        assert send.send("alice", {"k": 1}, "401", "401").result(timeout=30)
        assert recv.get_data("alice", "401", "401").result(timeout=30) == {
            "k": 1
        }
    finally:
        send.stop()
        recv.stop()


def test_chunk_streamed_send_roundtrip(monkeypatch):
    """Tensor frames above 2x the stripe size travel as KIND_CHUNKED main
    frame + sidecar chunk frames consumed in arrival order; content and a
    trailing small frame must both survive."""
    torch = pytest.importorskip("torch")
    from rayfed_amd.proxy.xfer import xfer_available

    if not xfer_available():
        pytest.skip("C++ transport extension not built")
    monkeypatch.setenv("RAYFED_SHM", "0")  # force the socket/payload route
    from rayfed_amd.proxy.xfer import XferReceiverService, XferSenderService

    addrs = make_addresses(["alice"])
    cfg = GrpcCrossSiloMessageConfig.from_dict(
        {"messages_max_size_in_bytes": 1 << 20}
    )
    recv = XferReceiverService(addrs["alice"], "alice", "j", cfg)
    send = XferSenderService(addrs, "alice", "j", cfg)
    try:
        t = torch.arange(3 << 20, dtype=torch.int16)  # 6 MiB > 2x 1 MiB
        obj = {"w": t, "tag": 5}
        assert send.send("alice", obj, "500", "500").result(timeout=60)
        out = recv.get_data("alice", "500", "500").result(timeout=60)
        assert out["tag"] == 5 and torch.equal(out["w"], t)
        assert send.send("alice", 99, "501", "501").result(timeout=30)
        assert recv.get_data("alice", "501", "501").result(timeout=30) == 99
    finally:
        send.stop()
        recv.stop()


def test_gil_switch_interval_tuned_and_restored():
    """fed.init lowers the GIL switch interval (hot-path wakeup latency);
    shutdown restores it; RAYFED_GIL_SWITCH_US=0 opts out."""
    import sys

    import rayfed_amd as fed

    base = sys.getswitchinterval()
    fed.init(addresses=make_addresses(["alice"]), party="alice",
             logging_level="warning")
    try:
        assert abs(sys.getswitchinterval() - 0.00025) < 1e-6
    finally:
        fed.shutdown()
    assert abs(sys.getswitchinterval() - base) < 1e-9


def test_chunk_stream_aborted_by_error_frame():
    """A chunk-streamed receive whose sender dies mid-stream: the error
    object substituted on the SAME seq ids must abort the chunk wait and
    surface to the consumer (never hang on missing chunks)."""
    from rayfed_amd._private import serialization
    from rayfed_amd.exceptions import FedRemoteError
    from rayfed_amd.proxy.grpc import frames
    from rayfed_amd.proxy.xfer import xfer_available

    if not xfer_available():
        pytest.skip("C++ transport extension not built")
    from rayfed_amd.proxy.xfer import XferReceiverService, XferSenderService

    addrs = make_addresses(["alice"])
    recv = XferReceiverService(addrs["alice"], "alice", "j", None)
    send = XferSenderService(addrs, "alice", "j", None)
    try:
        host, port = addrs["alice"].rsplit(":", 1)
        # Main frame announcing 3 chunks of a tensor payload...
        inner = frames.encode_frame_prefix(
            frames.KIND_TENSOR,
            {"job": "j", "up": "800", "down": "800", "skel": 4,
             "tensors": []},
        )
        meta = {"job": "j", "up": "800", "down": "800",
                "xk": 3, "xc": 1 << 20, "xlen": 3 << 20}
        main = [frames.encode_frame_prefix(frames.KIND_CHUNKED, meta), inner]
        code, _ = send._client_ctl.send(host, int(port), "800", "800",
                                        main, False, 30.0)
        assert code == 200
        # ...only chunk 0 ever arrives...
        code, _ = send._client_bulk.send(
            host, int(port), "800\x01x0", "800", [b"x" * (1 << 20)],
            False, 30.0, 2, True,
        )
        assert code == 200
        # ...then the sender's failure path substitutes an error object.
        err_frame = [
            frames.encode_frame_prefix(
                frames.KIND_ERROR, {"job": "j", "up": "800", "down": "800"}
            ),
            serialization.dumps(FedRemoteError("alice", None)),
        ]
        code, _ = send._client_ctl.send(host, int(port), "800", "800",
                                        err_frame, False, 30.0)
        assert code == 200
        with pytest.raises(FedRemoteError):
            recv.get_data("alice", "800", "800").result(timeout=60)
    finally:
        send.stop()
        recv.stop()


def test_chunk_stripes_env_knob(monkeypatch):
    """RAYFED_CHUNK_STRIPES controls per-chunk fanout, clamped to [1, 16]."""
    from rayfed_amd.proxy import xfer

    monkeypatch.delenv("RAYFED_CHUNK_STRIPES", raising=False)
    assert xfer._chunk_stripes() == 8  # measured-best default (TUNING.md)
    monkeypatch.setenv("RAYFED_CHUNK_STRIPES", "2")
    assert xfer._chunk_stripes() == 2
    monkeypatch.setenv("RAYFED_CHUNK_STRIPES", "0")
    assert xfer._chunk_stripes() == 1
    monkeypatch.setenv("RAYFED_CHUNK_STRIPES", "99")
    assert xfer._chunk_stripes() == 16
    monkeypatch.setenv("RAYFED_CHUNK_STRIPES", "not-a-number")
    assert xfer._chunk_stripes() == 8


def test_asyncio_request_timeout_cleans_pending():
    """A timed-out request must not leave its future in conn.pending."""
    import asyncio

    from rayfed_amd.proxy.tcp.tcp_proxy import _Connection

    async def scenario():
        async def handle(reader, writer):
            await reader.read(1 << 16)  # swallow the frame, never ack
            await asyncio.sleep(5)

        server = await asyncio.start_server(handle, "127.0.0.1", 0)
        port = server.sockets[0].getsockname()[1]
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        conn = _Connection(reader, writer)
        try:
            with pytest.raises(asyncio.TimeoutError):
                await conn.request(b"p", [b"x" * 64], timeout=0.2)
            assert not conn.pending  # slot reclaimed despite no ack
        finally:
            await conn.close()
            server.close()
            await server.wait_closed()

    asyncio.run(scenario())


def test_chunk_streamed_send_roundtrip_tls(monkeypatch, tmp_path):
    """The chunk-streamed path also rides mutual TLS: sidecar chunk frames
    stripe across parallel TLS connections and assemble server-side after
    decrypt, so big TLS frames get the same pipeline as plaintext."""
    torch = pytest.importorskip("torch")
    from rayfed_amd.proxy.xfer import xfer_available

    if not xfer_available():
        pytest.skip("C++ transport extension not built")
    monkeypatch.setenv("RAYFED_SHM", "0")  # force the socket/payload route
    import os as _os
    import sys as _sys

    _sys.path.insert(0, _os.path.join(_os.path.dirname(__file__), ".."))
    from tool.generate_tls_certs import generate

    tls = generate(str(tmp_path / "certs"))
    tls["target_name_override"] = "localhost"
    from rayfed_amd.proxy.xfer import XferReceiverService, XferSenderService

    addrs = make_addresses(["alice"])
    cfg = GrpcCrossSiloMessageConfig.from_dict(
        {"messages_max_size_in_bytes": 1 << 20}
    )
    recv = XferReceiverService(addrs["alice"], "alice", "j", cfg,
                               tls_config=tls)
    send = XferSenderService(addrs, "alice", "j", cfg, tls_config=tls)
    try:
        t = torch.arange(3 << 20, dtype=torch.int16)  # 6 MiB > 2x 1 MiB
        obj = {"w": t, "tag": 7}
        assert send.send("alice", obj, "900", "900").result(timeout=60)
        out = recv.get_data("alice", "900", "900").result(timeout=60)
        assert out["tag"] == 7 and torch.equal(out["w"], t)
        assert send.send("alice", 42, "901", "901").result(timeout=30)
        assert recv.get_data("alice", "901", "901").result(timeout=30) == 42
    finally:
        send.stop()
        recv.stop()


def test_duplicate_stripes_do_not_corrupt():
    """A retried frame re-sends stripes that already landed.  Duplicates
    must be drained (per-stripe seen flags), never double-counted into the
    assembly's byte total — double-counting would deliver the frame with
    the missing stripe's range zeroed.  Late duplicates of an already
    delivered frame are drained via the tombstone set."""
    import socket
    import struct

    from rayfed_amd._private import serialization
    from rayfed_amd.proxy.grpc import frames
    from rayfed_amd.proxy.xfer import XferReceiverService, xfer_available

    if not xfer_available():
        pytest.skip("C++ transport extension not built")

    addrs = make_addresses(["alice"])
    recv = XferReceiverService(addrs["alice"], "alice", "j", None)
    host, port = addrs["alice"].rsplit(":", 1)

    obj = {"blob": b"\xab" * (1 << 20), "tag": 123}
    body = frames.encode_frame(
        frames.KIND_PICKLE, {"job": "j", "up": "700", "down": "700"},
        serialization.dumps(obj),
    )
    half = len(body) // 2

    def send_stripe(idx, lo, hi, up="700", frame=None):
        frame = body if frame is None else frame
        names = bytes([2, 1, len(up), len(up)]) + b"j" + up.encode() * 2
        smeta = struct.pack("<IIQQ", idx, 2, lo, len(frame))
        payload = frame[lo:hi]
        total = 8 + len(names) + 24 + len(payload)
        with socket.create_connection((host, int(port)), timeout=10) as s:
            s.sendall(struct.pack("<QQ", total, 1) + names + smeta + payload)
            ack = b""
            while len(ack) < 14:
                chunk = s.recv(14 - len(ack))
                assert chunk, "server closed during ack"
                ack += chunk
            _len, _rid, code = struct.unpack("<IQH", ack)
            assert code == 200, code

    try:
        # stripe0, DUPLICATE stripe0, stripe1 — must still deliver intact.
        send_stripe(0, 0, half)
        send_stripe(0, 0, half)
        send_stripe(1, half, len(body))
        out = recv.get_data("alice", "700", "700").result(timeout=30)
        assert out == obj
        # Late duplicate after delivery: drained via tombstone, server fine.
        send_stripe(1, half, len(body))
        # A fresh normal frame still round-trips.
        body2 = frames.encode_frame(
            frames.KIND_PICKLE, {"job": "j", "up": "701", "down": "701"},
            serialization.dumps("after"),
        )
        h2 = len(body2) // 2
        send_stripe(0, 0, h2, up="701", frame=body2)
        send_stripe(1, h2, len(body2), up="701", frame=body2)
        assert recv.get_data("alice", "701", "701").result(timeout=30) == "after"
    finally:
        recv.stop()
