"""Per-job global context: seq-id allocator, runtime handles, shutdown flag.

Parity: /root/reference/fed/_private/global_context.py:22-121.

The deterministic monotonic ``next_seq_id`` is THE cross-party naming scheme:
every party runs the identical driver script, so the i-th fed call gets the
same id in every party and send/recv pairs rendezvous on ``(upstream_seq_id,
downstream_seq_id)`` without any negotiation (SURVEY.md §7 "hard parts").
Preserve this contract exactly — any divergence deadlocks the parties.
"""
from __future__ import annotations

import threading
from typing import Callable, Optional

from rayfed_amd.cleanup import CleanupManager
from rayfed_amd.runtime.executor import Executor


class GlobalContext:
    def __init__(
        self,
        job_name: str,
        current_party: str,
        sending_failure_handler: Optional[Callable[[Exception], None]] = None,
        exit_on_sending_failure: bool = False,
        continue_waiting_for_data_sending_on_error: bool = False,
    ):
        self._job_name = job_name
        self._current_party = current_party
        self._seq_count = 0
        self._seq_lock = threading.Lock()
        self._cleanup_manager = CleanupManager(
            current_party, self.acquire_shutdown_flag
        )
        self._executor = Executor()
        self._sending_failure_handler = sending_failure_handler
        self._exit_on_sending_failure = exit_on_sending_failure
        self._continue_waiting_for_data_sending_on_error = (
            continue_waiting_for_data_sending_on_error
        )
        self._atomic_shutdown_flag_lock = threading.Lock()
        self._atomic_shutdown_flag = True
        self._last_received_error: Optional[Exception] = None

    def next_seq_id(self) -> int:
        with self._seq_lock:
            self._seq_count += 1
            return self._seq_count

    def get_job_name(self) -> str:
        return self._job_name

    def get_current_party(self) -> str:
        return self._current_party

    def get_cleanup_manager(self) -> CleanupManager:
        return self._cleanup_manager

    def get_executor(self) -> Executor:
        return self._executor

    def get_sending_failure_handler(self):
        return self._sending_failure_handler

    def get_exit_on_sending_failure(self) -> bool:
        return self._exit_on_sending_failure

    def get_continue_waiting_for_data_sending_on_error(self) -> bool:
        return self._continue_waiting_for_data_sending_on_error

    def get_last_received_error(self) -> Optional[Exception]:
        return self._last_received_error

    def set_last_received_error(self, err: Exception) -> None:
        self._last_received_error = err

    def acquire_shutdown_flag(self) -> bool:
        """Return True exactly once — gates the failure-triggered shutdown.

        Parity: global_context.py:70-87 in the reference (once-only lock so
        concurrent failure paths trigger a single shutdown).
        """
        with self._atomic_shutdown_flag_lock:
            if self._atomic_shutdown_flag:
                self._atomic_shutdown_flag = False
                return True
            return False


_global_context: Optional[GlobalContext] = None
_global_context_lock = threading.Lock()


def init_global_context(
    current_party: str,
    job_name: str,
    sending_failure_handler: Optional[Callable[[Exception], None]] = None,
    exit_on_sending_failure: bool = False,
    continue_waiting_for_data_sending_on_error: bool = False,
) -> GlobalContext:
    global _global_context
    with _global_context_lock:
        if _global_context is None:
            _global_context = GlobalContext(
                job_name,
                current_party,
                sending_failure_handler=sending_failure_handler,
                exit_on_sending_failure=exit_on_sending_failure,
                continue_waiting_for_data_sending_on_error=(
                    continue_waiting_for_data_sending_on_error
                ),
            )
        return _global_context


def get_global_context() -> Optional[GlobalContext]:
    return _global_context


def clear_global_context(wait_for_sending: bool = False) -> None:
    global _global_context
    with _global_context_lock:
        if _global_context is not None:
            _global_context.get_cleanup_manager().stop(
                wait_for_sending=wait_for_sending
            )
            _global_context.get_executor().shutdown(wait=False)
            _global_context = None
