"""3-party integration (BASELINE config 5 topology: alice/bob/carol) +
combined sender-receiver proxy variant."""
import rayfed_amd as fed
from tests._util import run_parties


def _driver_three_party(party, addresses):
    fed.init(addresses=addresses, party=party, logging_level="warning")

    @fed.remote
    def make(v):
        return v

    @fed.remote
    def agg3(a, b, c):
        return a + b + c

    x = make.party("alice").remote(1)
    y = make.party("bob").remote(2)
    z = make.party("carol").remote(4)
    s = agg3.party("carol").remote(x, y, z)
    assert fed.get(s) == 7
    # Broadcast rule reaches BOTH other parties on fed.get of an owned obj.
    o = make.party(party).remote(10)
    assert fed.get(o) == 10
    fed.shutdown()


def test_three_parties():
    run_parties(_driver_three_party, parties=("alice", "bob", "carol"), timeout=120)


def _driver_three_party_ring(party, addresses):
    """Each party pushes to its ring successor — exercises concurrent
    pairwise channels among 3 receivers."""
    fed.init(addresses=addresses, party=party, logging_level="warning")
    parties = ["alice", "bob", "carol"]

    @fed.remote
    def make(tag):
        return tag

    @fed.remote
    def stamp(x, who):
        return f"{x}->{who}"

    outs = []
    for i, p in enumerate(parties):
        succ = parties[(i + 1) % 3]
        o = make.party(p).remote(p)
        outs.append(stamp.party(succ).remote(o, succ))
    vals = fed.get(outs)
    assert vals == ["alice->bob", "bob->carol", "carol->alice"]
    fed.shutdown()


def test_three_party_ring():
    run_parties(_driver_three_party_ring, parties=("alice", "bob", "carol"), timeout=120)


def _driver_combined_proxy(party, addresses):
    from rayfed_amd.proxy.tcp.combined import TcpSenderReceiverProxy

    fed.init(
        addresses=addresses,
        party=party,
        receiver_sender_proxy_cls=TcpSenderReceiverProxy,
        logging_level="warning",
    )

    @fed.remote
    def make():
        return 21

    @fed.remote
    def double(x):
        return x * 2

    o = make.party("alice").remote()
    r = double.party("bob").remote(o)
    assert fed.get(r) == 42
    fed.shutdown()


def test_combined_sender_receiver_proxy():
    run_parties(_driver_combined_proxy, timeout=90)
