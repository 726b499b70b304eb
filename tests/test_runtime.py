"""Substrate units: executor, actors, object refs, message queue, cleanup."""
import threading
import time

import pytest

from rayfed_amd._private.message_queue import MessageQueueManager
from rayfed_amd.runtime.executor import Executor
from rayfed_amd.runtime.object_ref import ObjectRef


def test_submit_and_result():
    ex = Executor()
    try:
        ref = ex.submit(lambda a, b: a + b, args=(1, 2))
        assert ref.result(timeout=10) == 3
    finally:
        ex.shutdown()


def test_submit_num_returns():
    ex = Executor()
    try:
        a, b = ex.submit(lambda: (1, 2), num_returns=2)
        assert a.result(10) == 1 and b.result(10) == 2
        bad = ex.submit(lambda: 5, num_returns=2)
        with pytest.raises(ValueError):
            bad[0].result(10)
    finally:
        ex.shutdown()


def test_task_exception_propagates():
    ex = Executor()
    try:
        ref = ex.submit(lambda: 1 / 0)
        with pytest.raises(ZeroDivisionError):
            ref.result(10)
    finally:
        ex.shutdown()


def test_actor_ordering():
    ex = Executor()
    try:

        class Acc:
            def __init__(self):
                self.log = []

            def slow(self):
                time.sleep(0.05)
                self.log.append("slow")
                return list(self.log)

            def fast(self):
                self.log.append("fast")
                return list(self.log)

        h = ex.create_actor(Acc)
        r1 = h.call("slow")
        r2 = h.call("fast")
        assert r1.result(10) == ["slow"]
        assert r2.result(10) == ["slow", "fast"]
    finally:
        ex.shutdown()


def test_actor_kill():
    ex = Executor()
    try:

        class A:
            def m(self):
                return 1

        h = ex.create_actor(A)
        assert h.call("m").result(10) == 1
        h.kill()
        ref = h.call("m")
        with pytest.raises(RuntimeError):
            ref.result(10)
    finally:
        ex.shutdown()


def test_object_ref_chain():
    ref = ObjectRef.from_value(2)
    doubled = ref.chain(lambda v: v * 2)
    assert doubled.result(5) == 4
    err = ObjectRef.from_exception(ValueError("x")).chain(lambda v: v)
    with pytest.raises(ValueError):
        err.result(5)


def test_message_queue_drain_on_stop():
    seen = []
    mq = MessageQueueManager(seen.append, "t")
    mq.start()
    for i in range(100):
        mq.append(i)
    mq.stop(wait_for_sending=True)
    assert seen == list(range(100))


def test_message_queue_abandon():
    gate = threading.Event()
    seen = []

    def handler(x):
        gate.wait(5)
        seen.append(x)

    mq = MessageQueueManager(handler, "t")
    mq.start()
    for i in range(50):
        mq.append(i)
    gate.set()
    mq.stop(wait_for_sending=False)
    # Abandoned: far fewer than 50 processed (at least the in-flight one).
    assert len(seen) < 50


def test_message_queue_is_event_driven():
    """Latency from append to handling must be far below the reference's
    0.1 s poll interval."""
    done = threading.Event()
    mq = MessageQueueManager(lambda x: done.set(), "t")
    mq.start()
    t0 = time.perf_counter()
    mq.append(1)
    assert done.wait(1.0)
    dt = time.perf_counter() - t0
    mq.stop()
    assert dt < 0.05, f"queue handling took {dt*1e3:.1f} ms — not event-driven"


def test_shutdown_flag_acquired_once():
    """Concurrent failure paths trigger exactly one shutdown
    (parity: reference global_context.py:70-87 / cleanup shutdown-once)."""
    from rayfed_amd._private.global_context import GlobalContext

    ctx = GlobalContext("j", "alice")
    try:
        import concurrent.futures as cf

        with cf.ThreadPoolExecutor(8) as pool:
            results = list(pool.map(lambda _: ctx.acquire_shutdown_flag(), range(32)))
        assert sum(results) == 1
    finally:
        ctx.get_cleanup_manager().stop(wait_for_sending=False)
        ctx.get_executor().shutdown(wait=False)
