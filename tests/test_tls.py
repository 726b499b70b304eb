"""Cross-party TLS with mutual auth (coverage parity: reference
test_enable_tls_across_parties.py, test_transport_proxy_tls.py)."""
import os
import sys

import pytest

import rayfed_amd as fed
from tests._util import run_parties

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from tool.generate_tls_certs import generate  # noqa: E402


@pytest.fixture(scope="module")
def tls_config(tmp_path_factory):
    cert_dir = str(tmp_path_factory.mktemp("certs"))
    paths = generate(cert_dir)
    paths["target_name_override"] = "localhost"
    return paths


def _driver_tls(party, addresses, tls_config):
    fed.init(addresses=addresses, party=party, tls_config=tls_config)

    @fed.remote
    def make(v):
        return v * 2

    @fed.remote
    def agg(a, b):
        return a + b

    x = make.party("alice").remote(1)
    y = make.party("bob").remote(2)
    s = agg.party("bob").remote(x, y)
    assert fed.get(s) == 6
    fed.shutdown()


def test_two_parties_over_tls(tls_config):
    run_parties(_driver_tls, args=(tls_config,), timeout=120)


def _driver_plaintext_to_tls_fails(party, addresses, tls_config):
    """A plaintext client cannot get an ack out of a TLS receiver."""
    if party != "alice":
        return
    fed.init(addresses=addresses, party=party, tls_config=tls_config)
    import socket

    from rayfed_amd.proxy.grpc import frames

    raw = frames.encode_frame(
        frames.KIND_PICKLE, {"job": "Anonymous_job", "up": "1", "down": "1"}, b"x"
    )
    host, port = addresses["alice"].rsplit(":", 1)
    try:
        with socket.create_connection((host, int(port)), timeout=5) as s:
            body = (1).to_bytes(8, "little") + raw
            s.sendall(len(body).to_bytes(8, "little") + body)
            s.settimeout(5)
            data = s.recv(4)
            # A TLS server must not complete our plaintext "request"; any
            # bytes it sends back are a TLS alert/handshake, never an ack.
            if data[:1] not in (b"", b"\x15", b"\x16"):
                fed.shutdown()
                sys.exit(9)
    except (ConnectionError, socket.timeout, OSError):
        pass
    fed.shutdown()
    sys.exit(0)


def test_plaintext_rejected_by_tls_receiver(tls_config):
    run_parties(
        _driver_plaintext_to_tls_fails,
        parties=("alice", "bob"),
        args=(tls_config,),
        timeout=60,
        expect_codes=[0, 0],
    )


def _driver_tls_bulk(party, addresses, tls_config):
    """A multi-MB numpy payload rides the TLS bulk path (striped when the
    C++ core is active), not just the inline small-frame exchange.  The
    same-host /dev/shm lane is disabled so the bytes really cross TLS."""
    import numpy as np

    os.environ["RAYFED_SHM"] = "0"
    fed.init(addresses=addresses, party=party, tls_config=tls_config)

    @fed.remote
    def make():
        rng = np.random.default_rng(7)
        return rng.standard_normal((4, 1 << 20)).astype(np.float32)  # 16 MiB

    @fed.remote
    def check(a):
        rng = np.random.default_rng(7)
        want = rng.standard_normal((4, 1 << 20)).astype(np.float32)
        assert np.array_equal(a, want)
        return float(a.sum())

    x = make.party("alice").remote()
    s = check.party("bob").remote(x)
    assert isinstance(fed.get(s), float)
    fed.shutdown()


def test_bulk_tensor_over_tls(tls_config):
    run_parties(_driver_tls_bulk, args=(tls_config,), timeout=180)
