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
    """A plaintext client cannot reach a TLS receiver."""
    if party != "alice":
        return
    fed.init(addresses=addresses, party=party, tls_config=tls_config)
    import grpc

    from rayfed_amd.proxy.grpc import frames

    channel = grpc.insecure_channel(addresses["alice"])
    stub = channel.unary_unary(
        frames.SEND_DATA_METHOD,
        request_serializer=frames.identity_serializer,
        response_deserializer=frames.identity_deserializer,
    )
    raw = frames.encode_frame(
        frames.KIND_PICKLE, {"job": "Anonymous_job", "up": "1", "down": "1"}, b"x"
    )
    try:
        stub(raw, timeout=5)
    except grpc.RpcError:
        fed.shutdown()
        sys.exit(0)
    fed.shutdown()
    sys.exit(9)


def test_plaintext_rejected_by_tls_receiver(tls_config):
    run_parties(
        _driver_plaintext_to_tls_fails,
        parties=("alice", "bob"),
        args=(tls_config,),
        timeout=60,
        expect_codes=[0, 0],
    )
