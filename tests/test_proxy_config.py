"""Proxy configuration plumb-through (coverage parity: reference
test_setup_proxy_actor.py / test_grpc_options_on_proxies.py — the per-job
cross-silo config must reach the live proxy objects)."""
import pytest

import rayfed_amd as fed
import rayfed_amd.proxy.barriers as barriers
from rayfed_amd.config import GrpcCrossSiloMessageConfig
from tests._util import run_parties


def _driver_config_reaches_proxy(party, addresses):
    fed.init(
        addresses=addresses,
        party=party,
        config={
            "cross_silo_comm": {
                "timeout_in_ms": 12345,
                "max_concurrency": 77,
                "send_resource_label": {"device": "gpu0"},
                "recv_resource_label": {"device": "gpu0"},
                "http_header": {"x-auth": "tok"},
                "messages_max_size_in_bytes": 5 * 1024 * 1024,
            }
        },
    )
    sender = barriers.get_service(barriers.sender_proxy_name())
    receiver = barriers.get_service(barriers.receiver_proxy_name())
    scfg = sender.proxy._proxy_config
    rcfg = receiver.proxy._proxy_config
    assert isinstance(scfg, GrpcCrossSiloMessageConfig)
    assert scfg.timeout_in_ms == 12345
    assert scfg.max_concurrency == 77
    assert scfg.send_resource_label == {"device": "gpu0"}
    assert rcfg.recv_resource_label == {"device": "gpu0"}
    assert scfg.http_header == {"x-auth": "tok"}
    assert rcfg.messages_max_size_in_bytes == 5 * 1024 * 1024

    # And the job still works under the custom config.
    @fed.remote
    def f(v):
        return v + 1

    @fed.remote
    def g(x):
        return x * 3

    o = f.party("alice").remote(1)
    r = g.party("bob").remote(o)
    assert fed.get(r) == 6
    fed.shutdown()


def test_cross_silo_config_reaches_proxies():
    run_parties(_driver_config_reaches_proxy)


def test_unknown_proxy_cls_signature_rejected():
    with pytest.raises(Exception):
        fed.init(
            addresses={"alice": "127.0.0.1:1"},
            party="alice",
            sender_proxy_cls=object,  # wrong signature -> loud failure
        )
    # cleanup any partial state
    try:
        fed.shutdown()
    except Exception:
        pass
