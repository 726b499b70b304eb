"""Concurrency stress: many interleaved cross-party objects, mixed payloads,
3 parties — shakes out races in the transport/mailbox/cleanup machinery."""
import random

import pytest

torch = pytest.importorskip("torch")

import rayfed_amd as fed  # noqa: E402
from tests._util import run_parties  # noqa: E402


def _driver_stress(party, addresses):
    fed.init(addresses=addresses, party=party, logging_level="warning")
    parties = sorted(addresses)
    rng = random.Random(1234)  # identical stream in every party

    @fed.remote
    def make(i, kind):
        if kind == 0:
            return i
        if kind == 1:
            return {"i": i, "blob": b"x" * rng_size(i)}
        return torch.full((257 + i % 513,), float(i))

    def rng_size(i):
        return 1000 + (i * 2654435761) % 50000

    @fed.remote
    def check(v, i, kind):
        if kind == 0:
            assert v == i
        elif kind == 1:
            assert v["i"] == i and len(v["blob"]) == rng_size(i)
        else:
            assert float(v[0]) == float(i) and v.numel() == 257 + i % 513
        return i

    outs = []
    for i in range(120):
        src = parties[rng.randrange(len(parties))]
        dst = parties[rng.randrange(len(parties))]
        kind = rng.randrange(3)
        o = make.party(src).remote(i, kind)
        outs.append(check.party(dst).remote(o, i, kind))
    vals = fed.get(outs)
    assert vals == list(range(120))
    fed.shutdown()


def test_stress_two_party():
    run_parties(_driver_stress, timeout=180)


def test_stress_three_party():
    run_parties(_driver_stress, parties=("alice", "bob", "carol"), timeout=240)
