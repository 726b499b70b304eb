"""End-to-end semantics under EVERY transport.

The integration suite normally runs on whatever transport auto-selection
picks (the C++ core when built).  The reference has one transport, so its
suite covers it by construction (/root/reference/fed/proxy/grpc/grpc_proxy.py);
here the same driver flows run under cpp, asyncio and gRPC to pin that all
three lanes carry identical semantics: argument passing, broadcast-on-get,
num_returns, nested containers, and cross-party error propagation.
"""
import os

import pytest

from tests._util import run_parties

TRANSPORTS = ["cpp", "asyncio", "grpc"]


def _init(party, addresses, transport):
    import rayfed_amd as fed

    kwargs = {}
    if transport == "grpc":
        os.environ["RAYFED_TRANSPORT"] = "asyncio"  # keep auto-select out
        from rayfed_amd.proxy.grpc.grpc_proxy import (
            GrpcReceiverProxy,
            GrpcSenderProxy,
        )

        kwargs = {
            "sender_proxy_cls": GrpcSenderProxy,
            "receiver_proxy_cls": GrpcReceiverProxy,
        }
    else:
        os.environ["RAYFED_TRANSPORT"] = transport
    fed.init(addresses=addresses, party=party, logging_level="warning",
             **kwargs)
    return fed


def _driver_semantics(party, addresses, transport):
    if transport == "cpp":
        from rayfed_amd.proxy.xfer import xfer_available

        if not xfer_available():
            return
    fed = _init(party, addresses, transport)

    @fed.remote
    def make(x):
        return {"v": x, "l": [x, x + 1]}

    @fed.remote
    def consume(d, bonus=0):
        return d["v"] + d["l"][1] + bonus

    @fed.remote
    def two():
        return 7, 8

    @fed.remote
    class Counter:
        def __init__(self):
            self.n = 0

        def add(self, k):
            self.n += k
            return self.n

    # Cross-party args in containers.
    o = make.party("alice").remote(10)
    r = consume.party("bob").remote(o, bonus=1)
    assert fed.get(r) == 22
    # num_returns + broadcast-on-get.
    a, b = two.party("alice").options(num_returns=2).remote()
    assert fed.get([a, b]) == [7, 8]
    # Ordered actor on the non-driver party.
    c = Counter.party("bob").remote()
    refs = [c.add.remote(1) for _ in range(5)]
    assert fed.get(refs[-1]) == 5
    fed.shutdown()


def _driver_error(party, addresses, transport):
    if transport == "cpp":
        from rayfed_amd.proxy.xfer import xfer_available

        if not xfer_available():
            return
    fed = _init(party, addresses, transport)
    from rayfed_amd.exceptions import FedRemoteError

    @fed.remote
    def boom():
        raise ValueError("boom")

    @fed.remote
    def use(x):
        return x

    o = boom.party("alice").remote()
    r = use.party("bob").remote(o)
    if party == "bob":
        with pytest.raises(FedRemoteError):
            fed.get(r)
    else:
        try:
            fed.get(r)
        except FedRemoteError:
            pass
    fed.shutdown()


@pytest.mark.parametrize("transport", TRANSPORTS)
def test_semantics_matrix(transport):
    run_parties(_driver_semantics, args=(transport,), timeout=120)


@pytest.mark.parametrize("transport", TRANSPORTS)
def test_error_matrix(transport):
    run_parties(_driver_error, args=(transport,), timeout=120)
