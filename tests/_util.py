"""Shared helpers for multi-party integration tests.

Pattern (mirrors the reference's test harness shape): each
party runs the *identical* driver function in its own process with
``fed.init`` on loopback addresses; the parent asserts exit codes.  The
transport is real gRPC over loopback — no mocks.
"""
from __future__ import annotations

import multiprocessing
import socket
from typing import Callable, Dict, Sequence

_mp = multiprocessing.get_context("fork")


def free_ports(n: int) -> list:
    socks = []
    ports = []
    for _ in range(n):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        socks.append(s)
        ports.append(s.getsockname()[1])
    for s in socks:
        s.close()
    return ports


def make_addresses(parties: Sequence[str]) -> Dict[str, str]:
    ports = free_ports(len(parties))
    return {p: f"127.0.0.1:{port}" for p, port in zip(parties, ports)}


def run_parties(
    target: Callable,
    parties: Sequence[str] = ("alice", "bob"),
    args: tuple = (),
    timeout: float = 90,
    expect_codes: Sequence[int] | None = None,
):
    """Spawn one process per party running ``target(party, addresses, *args)``;
    assert every exit code (default: all zero)."""
    addresses = make_addresses(parties)
    procs = [
        _mp.Process(target=target, args=(p, addresses) + args, name=f"party-{p}")
        for p in parties
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=timeout)
    codes = []
    for p in procs:
        if p.is_alive():
            p.terminate()
            p.join(5)
            raise AssertionError(f"{p.name} timed out after {timeout}s")
        codes.append(p.exitcode)
    expect = list(expect_codes) if expect_codes is not None else [0] * len(procs)
    assert codes == expect, f"party exit codes {codes}, expected {expect}"
    return codes
