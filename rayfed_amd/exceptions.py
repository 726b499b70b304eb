"""Cross-party error carriers.

Parity: /root/reference/fed/exceptions.py:16-25 (FedRemoteError semantics —
an error raised by a task in one party is delivered to every peer that waits
on that task's output, carrying the source party and, when the job config
allows it (``expose_error_trace``), the original cause).
"""
from __future__ import annotations


class FedRemoteError(Exception):
    """An exception raised inside a remote party's task, delivered cross-silo.

    The receiving party re-raises this from ``fed.get`` (or any barrier that
    consumes the poisoned object).  ``cause`` is the original exception when
    ``expose_error_trace`` is enabled in the job config, else ``None``.
    """

    def __init__(self, src_party: str, cause: Exception | None = None):
        self._src_party = src_party
        self._cause = cause
        super().__init__(src_party, cause)

    @property
    def src_party(self) -> str:
        return self._src_party

    @property
    def cause(self) -> Exception | None:
        return self._cause

    def __str__(self) -> str:
        msg = f"FedRemoteError occurred at {self._src_party}"
        if self._cause is not None:
            msg += f" caused by {self._cause!r}"
        return msg


class ShutdownError(Exception):
    """Raised when an operation is attempted on a shut-down fed runtime."""
