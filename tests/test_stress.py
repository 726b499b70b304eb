"""Concurrency stress: many interleaved cross-party objects, mixed payloads,
3 parties — shakes out races in the transport/mailbox/cleanup machinery."""
import random

import pytest

torch = pytest.importorskip("torch")

import rayfed_amd as fed  # noqa: E402
from tests._util import run_parties  # noqa: E402


def _driver_stress(party, addresses):
    fed.init(addresses=addresses, party=party, logging_level="warning")
    parties = sorted(addresses)
    rng = random.Random(1234)  # identical stream in every party

    @fed.remote
    def make(i, kind):
        if kind == 0:
            return i
        if kind == 1:
            return {"i": i, "blob": b"x" * rng_size(i)}
        return torch.full((257 + i % 513,), float(i))

    def rng_size(i):
        return 1000 + (i * 2654435761) % 50000

    @fed.remote
    def check(v, i, kind):
        if kind == 0:
            assert v == i
        elif kind == 1:
            assert v["i"] == i and len(v["blob"]) == rng_size(i)
        else:
            assert float(v[0]) == float(i) and v.numel() == 257 + i % 513
        return i

    outs = []
    for i in range(120):
        src = parties[rng.randrange(len(parties))]
        dst = parties[rng.randrange(len(parties))]
        kind = rng.randrange(3)
        o = make.party(src).remote(i, kind)
        outs.append(check.party(dst).remote(o, i, kind))
    vals = fed.get(outs)
    assert vals == list(range(120))
    fed.shutdown()


def test_stress_two_party():
    run_parties(_driver_stress, timeout=180)


def test_stress_three_party():
    run_parties(_driver_stress, parties=("alice", "bob", "carol"), timeout=240)


# ---------------------------------------------------------------- soak
def _driver_soak(party, addresses, rounds):
    """Sustained mixed-payload aggregate loop: the
    many_tiny_tasks shape with rotating payload kinds, rounds-deep."""
    fed.init(addresses=addresses, party=party, logging_level="warning")
    parties = sorted(addresses)

    @fed.remote
    class Actor:
        def run(self, i, kind):
            if kind == 0:
                return 1
            if kind == 1:
                return {"i": i, "pad": b"p" * (64 + i % 1024)}
            return torch.full((33,), float(i))

    @fed.remote
    class Agg:
        def agg(self, i, kind, *vals):
            if kind == 0:
                return sum(vals)
            if kind == 1:
                return sum(v["i"] for v in vals)
            return float(sum(float(v[0]) for v in vals))

    actors = [Actor.party(p).remote() for p in parties]
    agg = Agg.party(parties[0]).remote()
    for i in range(rounds):
        kind = i % 3
        vals = [a.run.remote(i, kind) for a in actors]
        r = fed.get(agg.agg.remote(i, kind, *vals))
        if kind == 0:
            assert r == len(parties)
        elif kind == 1:
            assert r == i * len(parties)
        else:
            assert r == float(i) * len(parties)
    # Mailboxes drained: every parked payload was consumed.
    from rayfed_amd.proxy import barriers

    recv = barriers._receiver_service
    sent = barriers._sender_service._get_stats()["send_op_count"]
    assert sent >= rounds  # every round pushed at least one object
    assert recv.proxy.received_op_count >= rounds - 1
    fed.shutdown()


def test_soak_10k_rounds():
    """10k mixed-payload rounds in CI — sustained-load regression for the
    transport/mailbox/cleanup machinery (builder soaks run 100k+ on GPU)."""
    run_parties(_driver_soak, args=(10_000,), timeout=600)


# ------------------------------------------------------- tamper over a socket
def test_checksum_tamper_over_real_socket():
    """Corrupt one payload byte on the wire path of the C++ transport (after
    encode, before the socket write): the consumer must get a checksum
    failure, never silent corruption — and the lane must keep working for
    the next good frame."""
    import zlib

    from rayfed_amd.ops import tensor_codec
    from rayfed_amd.proxy.grpc import frames
    from rayfed_amd.proxy.xfer import xfer_available

    if not xfer_available():
        pytest.skip("C++ transport extension not built")
    from rayfed_amd.proxy.xfer import XferReceiverService, XferSenderService
    from tests._util import make_addresses

    addrs = make_addresses(["alice"])
    recv = XferReceiverService(addrs["alice"], "alice", "j", None)
    send = XferSenderService(addrs, "alice", "j", None)
    try:
        t = torch.arange(1 << 18, dtype=torch.float32)
        extras, parts = tensor_codec.encode(t, None, shm=False)
        raw = bytes(parts[1])
        extras["tensors"][0]["crc32"] = zlib.crc32(raw) & 0xFFFFFFFF
        header = {"job": "j", "up": "9", "down": "9",
                  "skel": extras["skel"], "tensors": extras["tensors"]}
        prefix = frames.encode_frame_prefix(frames.KIND_TENSOR, header)
        tampered = bytearray(raw)
        tampered[12345] ^= 0x40
        host, port = addrs["alice"].rsplit(":", 1)
        code, _ = send._client_bulk.send(
            host, int(port), "9", "9",
            [prefix, bytes(parts[0]), bytes(tampered)], False, 30.0,
        )
        assert code == 200  # transport delivered; corruption is in payload
        with pytest.raises(ValueError, match="CRC mismatch"):
            recv.get_data("alice", "9", "9").result(timeout=30)
        # Lane still healthy afterwards.
        assert send.send("alice", {"ok": 1}, "10", "10").result(timeout=30)
        assert recv.get_data("alice", "10", "10").result(timeout=30) == {
            "ok": 1
        }
    finally:
        send.stop()
        recv.stop()


# ------------------------------------------- receiver crash mid-transfer
def test_receiver_crash_mid_push_escalates():
    """A peer dying mid-job: the pending push exhausts its retry budget,
    the cleanup manager substitutes the error and exit_on_sending_failure
    exits the surviving party with code 1 (reference
    test_exit_on_failure_sending semantics, but with a peer that WAS up)."""
    from tests._util import make_addresses

    addresses = make_addresses(["alice", "bob"])
    import multiprocessing

    _mp = multiprocessing.get_context("fork")

    def alice_main():
        import time as _time

        import rayfed_amd as _fed

        def handler(err):
            pass

        _fed.init(
            addresses=addresses, party="alice", logging_level="warning",
            sending_failure_handler=handler,
            config={"cross_silo_comm": {
                "exit_on_sending_failure": True,
                "timeout_in_ms": 4000,
                "grpc_retry_policy": {"maxAttempts": 2,
                                      "initialBackoff": "0.2s"},
            }},
        )

        @_fed.remote
        def ping_round(x):
            return x + 1

        @_fed.remote
        def big(i):
            return torch.ones(1 << 21)  # 8 MiB — rides the defer-ack lane

        @_fed.remote
        def sink(t):
            return int(t.numel())

        o = ping_round.party("alice").remote(1)
        r = ping_round.party("bob").remote(o)
        assert _fed.get(r) == 3
        # Handshake: bob dies only after this value reaches him, so the
        # crash always lands mid-job — never before alice's round-1 recv
        # completes (a peer dying pre-broadcast hangs the recv by design,
        # exactly as the reference would).
        go = ping_round.party("alice").remote(10)
        _fed.get(go)
        # Keep pushing until a send hits the dead peer; exit_on_sending_
        # failure then SIGINTs this process out of the loop.
        for i in range(20):
            o2 = big.party("alice").remote(i)
            sink.party("bob").remote(o2)
            _time.sleep(0.75)
        raise RuntimeError("sending failure never escalated")

    def bob_main():
        import os as _os

        import rayfed_amd as _fed

        _fed.init(addresses=addresses, party="bob", logging_level="warning")

        @_fed.remote
        def ping_round(x):
            return x + 1

        @_fed.remote
        def big(i):
            return torch.ones(1 << 21)

        @_fed.remote
        def sink(t):
            return int(t.numel())

        o = ping_round.party("alice").remote(1)
        r = ping_round.party("bob").remote(o)
        assert _fed.get(r) == 3
        go = ping_round.party("alice").remote(10)
        assert _fed.get(go) == 11  # alice's round-1 view is complete
        _os._exit(3)

    pa = _mp.Process(target=alice_main)
    pb = _mp.Process(target=bob_main)
    pa.start()
    pb.start()
    pb.join(timeout=60)
    pa.join(timeout=90)
    if pa.is_alive():
        pa.terminate()
        raise AssertionError("alice hung instead of exiting on send failure")
    assert pb.exitcode == 3
    assert pa.exitcode == 1  # unintended shutdown path
