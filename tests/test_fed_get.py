"""fed.get semantics: broadcast-on-owned, FedAvg-style loop, seq-id
determinism across re-init (coverage parity: reference test_fed_get.py,
test_repeat_init.py, test_reset_context.py, test_async_startup_2_clusters.py)."""
import time

import rayfed_amd as fed
from tests._util import run_parties


def _driver_fedavg_loop(party, addresses):
    """3-epoch mean-aggregation loop; both parties converge on [3, 6, 9]
    (parity with the reference's FedAvg-shaped test asserting weights)."""
    fed.init(addresses=addresses, party=party)

    @fed.remote
    class Trainer:
        def __init__(self, step):
            self.step = step
            self.w = 0

        def set_weights(self, w):
            self.w = w
            return w

        def train(self):
            return self.w + self.step

    @fed.remote
    def mean(a, b):
        return (a + b) / 2

    alice_t = Trainer.party("alice").remote(3)
    bob_t = Trainer.party("bob").remote(3)

    history = []
    for _epoch in range(3):
        wa = alice_t.train.remote()
        wb = bob_t.train.remote()
        avg = mean.party("alice").remote(wa, wb)
        w = fed.get(avg)
        history.append(w)
        alice_t.set_weights.remote(w)
        bob_t.set_weights.remote(w)
    assert history == [3, 6, 9], history
    fed.shutdown()


def test_fedavg_style_loop():
    run_parties(_driver_fedavg_loop)


def _driver_get_multiple(party, addresses):
    fed.init(addresses=addresses, party=party)

    @fed.remote
    def make(v):
        return v

    objs = [make.party("alice" if i % 2 == 0 else "bob").remote(i) for i in range(6)]
    vals = fed.get(objs)
    assert vals == list(range(6))
    # Second get of the same objects: served from cache / local refs, and
    # the broadcast dedup means no new sends.
    vals2 = fed.get(objs)
    assert vals2 == vals
    fed.shutdown()


def test_get_list_and_recache():
    run_parties(_driver_get_multiple)


def _driver_repeat_init(party, addresses):
    for round_i in range(2):
        fed.init(addresses=addresses, party=party)

        @fed.remote
        def make():
            return 5

        @fed.remote
        def double(x):
            return x * 2

        o = make.party("alice").remote()
        d = double.party("bob").remote(o)
        assert fed.get(d) == 10
        fed.shutdown()


def test_repeat_init_same_process():
    run_parties(_driver_repeat_init, timeout=120)


def _driver_seq_ids_reset(party, addresses):
    """Seq ids restart at 1 after re-init (parity: test_reset_context.py)."""
    from rayfed_amd._private.global_context import get_global_context

    fed.init(addresses=addresses, party=party)
    ids1 = [get_global_context().next_seq_id() for _ in range(3)]
    fed.shutdown()
    fed.init(addresses=addresses, party=party)
    ids2 = [get_global_context().next_seq_id() for _ in range(3)]
    fed.shutdown()
    assert ids1 == ids2 == [1, 2, 3]


def test_seq_ids_identical_across_reinit():
    run_parties(_driver_seq_ids_reset)


def _driver_staggered(party, addresses):
    """bob starts 5 s late; gRPC retry policy absorbs the gap
    (parity: reference test_async_startup_2_clusters.py)."""
    if party == "bob":
        time.sleep(5)
    fed.init(addresses=addresses, party=party)

    @fed.remote
    def make():
        return 1

    @fed.remote
    def bump(x):
        return x + 1

    o = make.party("alice").remote()
    r = bump.party("bob").remote(o)
    assert fed.get(r) == 2
    fed.shutdown()


def test_async_startup_two_parties():
    run_parties(_driver_staggered, timeout=120)


def _driver_ping(party, addresses):
    fed.init(
        addresses=addresses,
        party=party,
        config={"barrier_on_initializing": True},
    )

    @fed.remote
    def f(v):
        return v

    @fed.remote
    def join(a, b):
        return a + b

    # Cross-party join so neither driver can finish (and tear down its
    # receiver) before the other has passed its own ping barrier.
    a = f.party("alice").remote("po")
    b = f.party("bob").remote("ng")
    assert fed.get(join.party("bob").remote(a, b)) == "pong"
    fed.shutdown()


def test_ping_others_barrier():
    run_parties(_driver_ping, timeout=120)
