"""Job namespacing: per-job proxy names, wrong-job rejection end to end
(coverage parity: reference multi-jobs/test_multi_proxy_actor.py,
multi-jobs/test_ignore_other_job_msg.py)."""
import rayfed_amd as fed
import rayfed_amd.proxy.barriers as barriers
from tests._util import run_parties


def _driver_per_job_proxy_names(party, addresses):
    fed.init(
        addresses=addresses,
        party=party,
        job_name="job_abc",
        config={"cross_silo_comm": {"use_global_proxy": False}},
    )
    assert barriers.get_service("SenderProxy-job_abc") is not None
    assert barriers.get_service("ReceiverProxy-job_abc") is not None
    assert barriers.get_service("SenderProxy") is None

    @fed.remote
    def f(v):
        return v + 1

    @fed.remote
    def g(x):
        return x * 10

    o = f.party("alice").remote(1)
    r = g.party("bob").remote(o)
    assert fed.get(r) == 20
    fed.shutdown()


def test_per_job_proxy_names():
    run_parties(_driver_per_job_proxy_names)


def _driver_named_job(party, addresses):
    fed.init(addresses=addresses, party=party, job_name="job_xyz")

    @fed.remote
    def f():
        return "ok"

    @fed.remote
    def h(x):
        return x + "!"

    o = f.party("alice").remote()
    r = h.party("bob").remote(o)
    assert fed.get(r) == "ok!"
    fed.shutdown()


def test_named_job_roundtrip():
    run_parties(_driver_named_job)
