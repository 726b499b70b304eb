"""Transport-level tests: real gRPC through the proxy services in ONE process
(parity: reference test_transport_proxy.py:43-241 — n sends → n concurrent
get_datas, wrong-job rejection, stats counters)."""
import pytest

import rayfed_amd.proxy.barriers as barriers
from rayfed_amd._private.global_context import (
    clear_global_context,
    init_global_context,
)
from tests._util import make_addresses


@pytest.fixture(scope="session")
def _tls_conf(tmp_path_factory):
    import os
    import sys

    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    from tool.generate_tls_certs import generate

    conf = generate(str(tmp_path_factory.mktemp("tp-certs")))
    conf["target_name_override"] = "localhost"
    return conf


@pytest.fixture(params=["tcp", "grpc", "tcp-tls"])
def party_env(request, _tls_conf):
    tls = None
    if request.param == "tcp-tls":
        # Transport-level TLS with mutual auth (coverage parity:
        # reference test_transport_proxy_tls.py).
        tls = _tls_conf
    if request.param.startswith("tcp"):
        from rayfed_amd.proxy.tcp.tcp_proxy import TcpReceiverProxy, TcpSenderProxy

        sender_cls, receiver_cls = TcpSenderProxy, TcpReceiverProxy
    else:
        from rayfed_amd.proxy.grpc.grpc_proxy import (
            GrpcReceiverProxy,
            GrpcSenderProxy,
        )

        sender_cls, receiver_cls = GrpcSenderProxy, GrpcReceiverProxy
    addrs = make_addresses(["alice"])
    init_global_context(current_party="alice", job_name="test_job")
    receiver = barriers.start_receiver_proxy(
        addrs, "alice", job_name="test_job", tls_config=tls,
        proxy_cls=receiver_cls, proxy_config=None,
    )
    sender = barriers.start_sender_proxy(
        addrs, "alice", job_name="test_job", tls_config=tls,
        proxy_cls=sender_cls, proxy_config=None,
    )
    yield request.param, addrs, sender, receiver
    clear_global_context()
    barriers._cleanup_proxies()


def test_n_to_1_send_recv(party_env):
    kind, addrs, sender, receiver = party_env
    n = 10
    sends = [
        barriers.send("alice", {"i": i}, f"up{i}", f"down{i}") for i in range(n)
    ]
    for f in sends:
        assert f.result(timeout=20) is True
    # Readers arrive after delivery: parked payloads resolve immediately.
    for i in range(n):
        ref = barriers.recv("alice", "alice", f"up{i}", f"down{i}")
        assert ref.result(timeout=20) == {"i": i}
    assert sender._get_stats()["send_op_count"] == n
    assert receiver._get_stats()["receive_op_count"] == n


def test_reader_before_sender(party_env):
    kind, addrs, sender, receiver = party_env
    ref = barriers.recv("alice", "alice", "late", "late")
    assert not ref.done()
    barriers.send("alice", [1, 2, 3], "late", "late").result(timeout=20)
    assert ref.result(timeout=20) == [1, 2, 3]


def test_wrong_job_name_rejected_with_417(party_env):
    """A raw frame with a mismatched job name gets code 417
    (parity: reference multi-jobs/test_ignore_other_job_msg.py)."""
    kind, addrs, _, _ = party_env
    from rayfed_amd.proxy.grpc import frames

    raw = frames.encode_frame(
        frames.KIND_PICKLE,
        {"job": "SOME_OTHER_JOB", "up": "1", "down": "1"},
        b"payload",
    )
    if kind == "tcp-tls":
        pytest.skip("raw-plaintext probe not applicable to the TLS listener")
    if kind == "grpc":
        import grpc

        channel = grpc.insecure_channel(addrs["alice"])
        stub = channel.unary_unary(
            frames.SEND_DATA_METHOD,
            request_serializer=frames.identity_serializer,
            response_deserializer=frames.identity_deserializer,
        )
        resp = frames.decode_response(stub(raw, timeout=10))
        channel.close()
    else:
        import socket

        import msgpack

        host, port = addrs["alice"].rsplit(":", 1)
        with socket.create_connection((host, int(port)), timeout=10) as s:
            body = (1).to_bytes(8, "little") + raw
            s.sendall(len(body).to_bytes(8, "little") + body)
            n = int.from_bytes(_read_n(s, 4), "little")
            resp = msgpack.unpackb(_read_n(s, n), raw=False)
    assert resp["code"] == 417


def _read_n(sock, n):
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("eof")
        buf += chunk
    return buf


def test_large_payload(party_env):
    kind, addrs, _, _ = party_env
    blob = b"z" * (8 * 1024 * 1024)
    barriers.send("alice", blob, "big", "big").result(timeout=30)
    ref = barriers.recv("alice", "alice", "big", "big")
    assert ref.result(timeout=30) == blob


def test_tensor_payload_over_wire(party_env):
    torch = pytest.importorskip("torch")
    kind, addrs, _, _ = party_env
    t = torch.randn(1000, dtype=torch.float32)
    barriers.send("alice", {"w": t}, "tens", "tens").result(timeout=30)
    out = barriers.recv("alice", "alice", "tens", "tens").result(timeout=30)
    assert torch.equal(out["w"], t)


def test_proxy_naming():
    assert barriers.sender_proxy_name("j", use_global_proxy=True) == "SenderProxy"
    assert barriers.sender_proxy_name("j", use_global_proxy=False) == "SenderProxy-j"
    assert (
        barriers.receiver_proxy_name("j", use_global_proxy=False)
        == "ReceiverProxy-j"
    )
