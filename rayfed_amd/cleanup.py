"""CleanupManager — send-flush on exit + cross-party failure propagation.

Parity: /root/reference/fed/cleanup.py:46-203.  Behavior pinned by the
reference's ``test_cross_silo_error.py`` suite:

- every cross-party send is ack-tracked asynchronously (data queue);
- when a send fails (the producing task raised, or the transport gave up),
  a ``FedRemoteError`` is sent **to the same seq ids** so the peer's pending
  recv unblocks with the error instead of hanging;
- with ``exit_on_sending_failure`` the party then shuts itself down via a
  once-only SIGINT to the main thread;
- stop order on shutdown: data queue drains before the error queue so error
  objects (which unblock the peer) always go out last.

Redesign vs reference: queues are event-driven (no 0.1 s polls) and the send
"task result" is a ``concurrent.futures.Future`` from the in-process sender
proxy, not a Ray ObjectRef.
"""
from __future__ import annotations

import logging
import os
import signal
import threading
from concurrent.futures import Future
from typing import Callable, Optional

from rayfed_amd._private.message_queue import MessageQueueManager

logger = logging.getLogger(__name__)


class _PendingSend:
    __slots__ = ("future", "dest_party", "upstream_seq_id", "downstream_seq_id")

    def __init__(self, future, dest_party, upstream_seq_id, downstream_seq_id):
        self.future = future
        self.dest_party = dest_party
        self.upstream_seq_id = upstream_seq_id
        self.downstream_seq_id = downstream_seq_id


class CleanupManager:
    def __init__(
        self,
        current_party: str,
        acquire_shutdown_flag: Callable[[], bool],
    ):
        self._current_party = current_party
        self._acquire_shutdown_flag = acquire_shutdown_flag
        self._sending_data_q = MessageQueueManager(
            self._process_data_sending_task_return, thread_name="DataSendingQueueThread"
        )
        self._sending_error_q = MessageQueueManager(
            self._process_error_sending_task_return, thread_name="ErrorSendingQueueThread"
        )
        self._monitor_thread: Optional[threading.Thread] = None
        self.exit_on_sending_failure = False
        self.expose_error_trace = False
        self.continue_waiting_for_data_sending_on_error = False

    def start(
        self,
        exit_on_sending_failure: bool = False,
        expose_error_trace: bool = False,
        continue_waiting_for_data_sending_on_error: bool = False,
    ) -> None:
        self.exit_on_sending_failure = exit_on_sending_failure
        self.expose_error_trace = expose_error_trace
        self.continue_waiting_for_data_sending_on_error = (
            continue_waiting_for_data_sending_on_error
        )
        self._sending_data_q.start()
        self._sending_error_q.start()
        logger.debug("CleanupManager started.")

    def stop(self, wait_for_sending: bool = True) -> None:
        """Drain (or abandon) queues.  Data queue stops before error queue —
        parity with cleanup.py:71-76 in the reference."""
        self._sending_data_q.stop(wait_for_sending=wait_for_sending)
        self._sending_error_q.stop(wait_for_sending=wait_for_sending)

    def push_to_sending(
        self,
        send_future: Future,
        dest_party: str,
        upstream_seq_id,
        downstream_seq_id,
        is_error: bool = False,
    ) -> None:
        """Track an in-flight cross-party send for ack / failure handling.

        Parity: cleanup.py:78-107 in the reference.
        """
        item = _PendingSend(send_future, dest_party, upstream_seq_id, downstream_seq_id)
        if is_error:
            self._sending_error_q.append(item)
        else:
            self._sending_data_q.append(item)

    # -- queue handlers -------------------------------------------------------
    def _process_data_sending_task_return(self, item: _PendingSend) -> bool:
        try:
            res = item.future.result()
            if res is False:
                raise RuntimeError(
                    f"send to {item.dest_party} returned a failure ack"
                )
            return True
        except Exception as e:  # noqa: BLE001
            logger.warning(
                "Failed to send %s/%s to %s: %r",
                item.upstream_seq_id,
                item.downstream_seq_id,
                item.dest_party,
                e,
            )
            from rayfed_amd.exceptions import FedRemoteError
            from rayfed_amd.proxy import barriers

            # Send an error object on the SAME seq ids so the peer's pending
            # recv resolves with the failure instead of hanging forever
            # (reference cleanup.py:160-172).
            cause = e if self.expose_error_trace else None
            error = FedRemoteError(self._current_party, cause)
            try:
                barriers.send(
                    dest_party=item.dest_party,
                    data=error,
                    upstream_seq_id=item.upstream_seq_id,
                    downstream_seq_id=item.downstream_seq_id,
                    is_error=True,
                )
            except Exception:  # noqa: BLE001
                logger.exception("failed to enqueue error-send to %s", item.dest_party)

            if self.exit_on_sending_failure:
                self._signal_exit()
                if not self.continue_waiting_for_data_sending_on_error:
                    # Abandon the remaining queued sends (asynchronously: the
                    # queue thread itself must not join itself).
                    self._sending_data_q.abandon()
            return False

    def _process_error_sending_task_return(self, item: _PendingSend) -> bool:
        try:
            item.future.result()
        except Exception:  # noqa: BLE001
            logger.exception(
                "Failed to send error object to %s for %s/%s",
                item.dest_party,
                item.upstream_seq_id,
                item.downstream_seq_id,
            )
        return True

    # -- failure-triggered shutdown -------------------------------------------
    def _signal_exit(self) -> None:
        """Once-only: interrupt the main thread so it runs the unintended-exit
        shutdown path (reference cleanup.py:112-128)."""
        if not self._acquire_shutdown_flag():
            return
        logger.warning("Signal SIGINT to exit on sending failure.")
        os.kill(os.getpid(), signal.SIGINT)
