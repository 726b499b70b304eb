"""API-level units: decorator validation, init validation, kill,
listening-address conflicts (coverage parity: reference test_api.py,
test_listening_address.py, test_kill)."""
import multiprocessing

import pytest

import rayfed_amd as fed
from tests._util import make_addresses, run_parties

_mp = multiprocessing.get_context("fork")


def test_remote_without_party_raises():
    @fed.remote
    def f():
        return 1

    with pytest.raises(ValueError, match="party"):
        f.remote()

    @fed.remote
    class A:
        pass

    with pytest.raises(ValueError, match="party"):
        A.remote()


def test_remote_rejects_non_callable():
    with pytest.raises(TypeError):
        fed.remote(42)


def test_init_validation():
    with pytest.raises(AssertionError):
        fed.init(addresses=None, party="alice")
    with pytest.raises(AssertionError):
        fed.init(addresses={"alice": "127.0.0.1:1"}, party=None)
    with pytest.raises(AssertionError):
        fed.init(addresses={"alice": "127.0.0.1:1"}, party="carol")
    with pytest.raises(ValueError):
        fed.init(addresses={"alice": "badaddress"}, party="alice")


def _driver_port_in_use(party, addresses):
    import socket

    # Occupy the party's own port first: receiver bind must fail loudly.
    s = socket.socket()
    host, port = addresses[party].split(":")
    s.bind((host, int(port)))
    s.listen(1)
    try:
        fed.init(addresses=addresses, party=party)
    except AssertionError:
        import sys

        sys.exit(7)  # expected path
    finally:
        s.close()


def test_listening_address_in_use():
    addresses = make_addresses(["alice"])
    p = _mp.Process(target=_driver_port_in_use, args=("alice", addresses))
    p.start()
    p.join(timeout=60)
    assert p.exitcode == 7


def _driver_kill(party, addresses):
    # Short transport deadline: the post-kill broadcast targets a peer that
    # may already have exited; shutdown's drain must not wait the default 60 s.
    fed.init(
        addresses=addresses,
        party=party,
        config={"cross_silo_comm": {"timeout_in_ms": 5000}},
    )

    @fed.remote
    class A:
        def ping(self):
            return "alive"

    a = A.party("alice").remote()
    r = a.ping.remote()
    if party == "alice":
        assert fed.get(r) == "alive"
    fed.kill(a)
    if party == "alice":
        r2 = a.ping.remote()
        with pytest.raises(Exception):
            fed.get(r2)
    fed.shutdown()


def test_kill_actor():
    run_parties(_driver_kill)


def _driver_local_get_of_plain_ref(party, addresses):
    fed.init(addresses=addresses, party=party)

    @fed.remote
    def f():
        return 5

    obj = f.party(party).remote()
    ref = obj.get_ray_object_ref()
    assert fed.get(ref) == 5  # plain ObjectRef passthrough
    fed.shutdown()


def test_get_plain_object_ref():
    run_parties(_driver_local_get_of_plain_ref, parties=("alice",))


def test_examples_demo_runs():
    """The README-demo example (examples/demo.py) runs end to end — the
    reference's README.md:124-168 flow on this engine."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = os.path.join(repo, "examples", "demo.py")
    pb = subprocess.Popen([sys.executable, script, "bob"],
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                          text=True, cwd=repo)
    pa = subprocess.run([sys.executable, script, "alice"],
                        capture_output=True, text=True, timeout=120, cwd=repo)
    out_b, _ = pb.communicate(timeout=120)
    assert pa.returncode == 0, pa.stdout + pa.stderr
    assert pb.returncode == 0, out_b
    assert "The result in party alice is 5" in pa.stdout
    assert "The result in party bob is 5" in out_b


def _driver_stats(party, addresses):
    import rayfed_amd as fed

    fed.init(addresses=addresses, party=party, logging_level="warning")

    @fed.remote
    def f(x):
        return x * 2

    o = f.party("alice").remote(3)
    r = f.party("bob").remote(o)
    assert fed.get(r) == 12
    s = fed.stats()
    assert s["send"]["send_op_count"] >= 1
    assert "edges" in s["send"]
    assert s["recv"]["receive_op_count"] >= 1
    fed.shutdown()


def test_stats_accessor():
    """fed.stats() exposes the per-edge transfer counters ([new] API)."""
    from tests._util import run_parties

    run_parties(_driver_stats)


def test_examples_secure_agg_runs():
    """The pairwise-masked secure-aggregation example runs end to end and
    the masks cancel exactly."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = os.path.join(repo, "examples", "secure_agg.py")
    pb = subprocess.Popen([sys.executable, script, "bob"],
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                          text=True, cwd=repo)
    pa = subprocess.run([sys.executable, script, "alice"],
                        capture_output=True, text=True, timeout=120, cwd=repo)
    out_b, _ = pb.communicate(timeout=120)
    assert pa.returncode == 0, pa.stdout + pa.stderr
    assert pb.returncode == 0, out_b
    assert "masked aggregate mean" in pa.stdout


def test_doctor_preflight():
    """`python -m rayfed_amd.doctor` passes on this box (required checks)."""
    import subprocess
    import sys

    proc = subprocess.run(
        [sys.executable, "-m", "rayfed_amd.doctor"],
        capture_output=True, text=True, timeout=120,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "cpp transport" in proc.stdout
    # Required rows must all be ok; warns are fine on CPU-only boxes.
    for line in proc.stdout.splitlines():
        assert not line.startswith("[FAIL]"), line


def test_doctor_run_checks_shape():
    from rayfed_amd import doctor

    rows = doctor.run_checks()
    names = {r[0] for r in rows}
    assert {"python", "torch", "cpp transport", "hip kernels", "gpu"} <= names
    for name, required, ok, detail in rows:
        assert isinstance(required, bool) and isinstance(ok, bool)
        assert isinstance(detail, str) and detail
