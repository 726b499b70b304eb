"""Failure-path: cross-party error propagation
(coverage parity: reference test_cross_silo_error.py:85-308,
test_exit_on_failure_sending.py)."""
import multiprocessing

import pytest

import rayfed_amd as fed
from rayfed_amd.exceptions import FedRemoteError
from tests._util import make_addresses, run_parties

_mp = multiprocessing.get_context("fork")


class MyError(Exception):
    pass


def _driver_task_error(party, addresses, expose):
    fed.init(
        addresses=addresses,
        party=party,
        config={"cross_silo_comm": {"expose_error_trace": expose}},
    )

    @fed.remote
    def boom():
        raise MyError("boom in alice")

    @fed.remote
    def consume(x):
        return x

    o = boom.party("alice").remote()
    c = consume.party("bob").remote(o)
    if party == "bob":
        with pytest.raises(FedRemoteError) as ei:
            fed.get(c)
        err = ei.value
        assert err.src_party == "alice"
        if expose:
            assert isinstance(err.cause, MyError)
        else:
            assert err.cause is None
    else:
        # alice's own get of the poisoned local object raises the original.
        with pytest.raises(MyError):
            fed.get(o)
    fed.shutdown()


def test_remote_task_error_reaches_peer_without_trace():
    run_parties(_driver_task_error, args=(False,))


def test_remote_task_error_reaches_peer_with_trace():
    run_parties(_driver_task_error, args=(True,))


def _driver_actor_error(party, addresses):
    fed.init(
        addresses=addresses,
        party=party,
        config={"cross_silo_comm": {"expose_error_trace": True}},
    )

    @fed.remote
    class Bad:
        def fail(self):
            raise MyError("actor failure")

    @fed.remote
    def consume(x):
        return x

    b = Bad.party("alice").remote()
    o = b.fail.remote()
    c = consume.party("bob").remote(o)
    if party == "bob":
        with pytest.raises(FedRemoteError):
            fed.get(c)
    else:
        with pytest.raises(MyError):
            fed.get(o)
    fed.shutdown()


def test_actor_method_error_reaches_peer():
    run_parties(_driver_actor_error)


def _driver_exit_on_sending_failure(party, addresses, q):
    """alice sends to a bob that never starts: retries exhaust, the failure
    handler runs, and the process exits 1 (parity: reference
    test_exit_on_failure_sending.py)."""
    if party == "bob":
        return  # bob never comes up

    def handler(err):
        q.put("handler_called")

    fed.init(
        addresses=addresses,
        party=party,
        config={
            "cross_silo_comm": {
                "exit_on_sending_failure": True,
                "timeout_in_ms": 5000,
                "grpc_retry_policy": {
                    "maxAttempts": 2,
                    "initialBackoff": "1s",
                    "maxBackoff": "1s",
                    "backoffMultiplier": 1,
                    "retryableStatusCodes": ["UNAVAILABLE"],
                },
            }
        },
        sending_failure_handler=handler,
    )

    @fed.remote
    def make():
        return 123

    @fed.remote
    def consume(x):
        return x

    o = make.party("alice").remote()
    consume.party("bob").remote(o)  # triggers the doomed send
    import time

    time.sleep(60)  # the SIGINT from the cleanup manager interrupts this
    fed.shutdown()


def test_exit_on_sending_failure_calls_handler_and_exits_1():
    addresses = make_addresses(["alice", "bob"])
    q = _mp.Queue()
    p = _mp.Process(
        target=_driver_exit_on_sending_failure, args=("alice", addresses, q)
    )
    p.start()
    p.join(timeout=90)
    assert not p.is_alive(), "alice did not exit"
    assert p.exitcode == 1
    assert q.get(timeout=5) == "handler_called"


def _driver_shutdown_flushes_sends(party, addresses):
    """A send still in flight at shutdown is delivered (drain-before-stop).

    alice shuts down while its 1 s producing task is still materializing the
    cross-party send; the flush-on-exit drain must deliver it.  bob reads the
    result from its local ref (not fed.get) so no broadcast targets the
    already-exited alice.
    """
    fed.init(addresses=addresses, party=party)

    @fed.remote
    def make():
        import time

        time.sleep(1.0)  # still materializing when shutdown starts
        return 9

    @fed.remote
    def consume(x):
        return x

    o = make.party("alice").remote()
    c = consume.party("bob").remote(o)
    if party == "bob":
        assert c.get_ray_object_ref().result(timeout=60) == 9
    fed.shutdown()


def test_shutdown_waits_for_pending_sends():
    run_parties(_driver_shutdown_flushes_sends)
