"""Event-driven message queue thread.

Parity: /root/reference/fed/_private/message_queue.py:40-103 — same API
(``start``/``append``/``appendleft``/``stop(wait)``, refuse stop from own
thread) but **event-driven**: a ``queue.Queue`` blocking get replaces the
reference's 0.1 s sleep-poll loop (message_queue.py:46), which is one of the
dominant latency terms in the reference's tiny-task benchmark (SURVEY.md
§3.5).  Here an enqueued send is picked up in microseconds.
"""
from __future__ import annotations

import logging
import queue
import threading
from typing import Any, Callable, Optional

logger = logging.getLogger(__name__)

_STOP = object()


class MessageQueueManager:
    def __init__(self, msg_handler: Callable[[Any], bool], thread_name: str = "mq"):
        self._msg_handler = msg_handler
        self._thread_name = thread_name
        self._queue: "queue.Queue" = queue.Queue()
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()
        self._abandon = threading.Event()

    def start(self) -> None:
        with self._lock:
            if self._thread is not None and self._thread.is_alive():
                return
            self._abandon.clear()
            self._thread = threading.Thread(
                target=self._loop, name=self._thread_name, daemon=True
            )
            self._thread.start()

    def _loop(self) -> None:
        while True:
            item = self._queue.get()
            if item is _STOP or self._abandon.is_set():
                break
            try:
                self._msg_handler(item)
            except Exception:  # noqa: BLE001 - a handler bug must not kill the loop
                logger.exception("message handler failed; continuing")

    def append(self, item: Any) -> None:
        self._queue.put(item)

    def appendleft(self, item: Any) -> None:
        # queue.Queue has no appendleft; emulate with an internal deque poke.
        with self._queue.mutex:
            self._queue.queue.appendleft(item)
            self._queue.not_empty.notify()

    def is_started(self) -> bool:
        return self._thread is not None and self._thread.is_alive()

    def abandon(self) -> None:
        """Drop whatever is still queued without joining the thread — safe to
        call from the queue's own handler (``stop(False)`` is not: it joins)."""
        self._abandon.set()

    def stop(self, wait_for_sending: bool = True) -> None:
        """Stop the polling thread.

        ``wait_for_sending=True`` drains every queued message first;
        ``False`` abandons whatever is still queued.  Never call from the
        queue's own thread (deadlock) — mirrored from the reference
        (message_queue.py:84-90).
        """
        thread = self._thread
        if thread is None or not thread.is_alive():
            return
        if threading.current_thread() is thread:
            logger.error(
                "cannot stop message queue %s from its own thread",
                self._thread_name,
            )
            return
        if not wait_for_sending:
            self._abandon.set()
            # Unblock the thread promptly even if the queue is empty.
        self._queue.put(_STOP)
        thread.join()
        self._thread = None
