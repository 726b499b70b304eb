"""2-party integration: basic cross-party object passing, actors, dedup,
multi-returns, containers (coverage parity: reference
test_basic_pass_fed_objects.py, test_cache_fed_objects.py, test_options.py,
test_pass_fed_objects_in_containers_in_*.py)."""
import rayfed_amd as fed
from tests._util import run_parties


def _driver_pass_objects(party, addresses):
    fed.init(addresses=addresses, party=party)

    @fed.remote
    def produce():
        return {"v": 41}

    @fed.remote
    def consume(x):
        return x["v"] + 1

    o = produce.party("alice").remote()
    c = consume.party("bob").remote(o)
    assert fed.get(c) == 42
    fed.shutdown()


def test_pass_fed_object_across_parties():
    run_parties(_driver_pass_objects)


def _driver_actors(party, addresses):
    fed.init(addresses=addresses, party=party)

    @fed.remote
    class Counter:
        def __init__(self, start):
            self.v = start

        def add(self, n):
            self.v += n
            return self.v

    a = Counter.party("alice").remote(10)
    b = Counter.party("bob").remote(100)
    va = a.add.remote(1)
    vb = b.add.remote(2)

    @fed.remote
    def agg(x, y):
        return x + y

    s = agg.party("bob").remote(va, vb)
    assert fed.get(s) == 11 + 102
    # Ordered actor semantics: second call sees first call's state.
    va2 = a.add.remote(1)
    assert fed.get(va2) == 12
    fed.shutdown()


def test_actors_across_parties():
    run_parties(_driver_actors)


def _driver_dedup(party, addresses):
    """A twice-consumed cross-party object is sent exactly once
    (parity: reference test_cache_fed_objects.py via proxy stats)."""
    import rayfed_amd.proxy.barriers as barriers

    fed.init(addresses=addresses, party=party)

    @fed.remote
    def produce():
        return 7

    @fed.remote
    def consume2(x, y):
        return x + y

    o = produce.party("alice").remote()
    c1 = consume2.party("bob").remote(o, o)
    c2 = consume2.party("bob").remote(o, o)
    assert fed.get(c1) == 14
    assert fed.get(c2) == 14

    sender = barriers.get_service(barriers.sender_proxy_name())
    receiver = barriers.get_service(barriers.receiver_proxy_name())
    if party == "alice":
        # One send for o (dedup), none for c1/c2 results until fed.get
        # broadcast (c1, c2 are owned by bob). fed.get(c1/c2) on alice is a
        # recv; alice sends only `o` once.
        assert sender._get_stats()["send_op_count"] == 1
    else:
        # bob receives `o` once; sends c1 and c2 broadcast once each.
        assert receiver.proxy.received_op_count == 1
        assert sender._get_stats()["send_op_count"] == 2
    fed.shutdown()


def test_cross_party_object_sent_once():
    run_parties(_driver_dedup)


def _driver_num_returns(party, addresses):
    fed.init(addresses=addresses, party=party)

    @fed.remote
    def two():
        return 1, 2

    a, b = two.party("alice").options(num_returns=2).remote()
    assert fed.get(a) == 1
    assert fed.get(b) == 2

    @fed.remote
    def plus(x, y):
        return x + y

    s = plus.party("bob").remote(a, b)
    assert fed.get(s) == 3
    fed.shutdown()


def test_num_returns_two():
    run_parties(_driver_num_returns)


def _driver_containers(party, addresses):
    fed.init(addresses=addresses, party=party)

    @fed.remote
    def produce(n):
        return n * 10

    @fed.remote
    def consume(container):
        (a, b), meta = container["pair"], container["meta"]
        return a + b + meta["bias"]

    x = produce.party("alice").remote(1)
    y = produce.party("bob").remote(2)
    c = consume.party("bob").remote({"pair": (x, y), "meta": {"bias": 5}})
    assert fed.get(c) == 10 + 20 + 5
    fed.shutdown()


def test_fed_objects_nested_in_containers():
    run_parties(_driver_containers)


def _driver_actor_containers(party, addresses):
    fed.init(addresses=addresses, party=party)

    @fed.remote
    class Holder:
        def keep(self, blob):
            self.blob = blob
            return sum(blob["vals"])

    @fed.remote
    def make(v):
        return v

    h = Holder.party("bob").remote()
    a = make.party("alice").remote(3)
    b = make.party("bob").remote(4)
    r = h.keep.remote({"vals": [a, b]})
    assert fed.get(r) == 7
    fed.shutdown()


def test_fed_objects_in_containers_to_actor():
    run_parties(_driver_actor_containers)
