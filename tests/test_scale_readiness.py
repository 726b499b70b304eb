"""8-GPU readiness, CPU-proven.

The driver's round-end scaling run launches bench.py under torchrun with
one rank per GPU.  These tests execute the EXACT launch shape on CPU
(gloo) so a broken rendezvous, lane split, or group bootstrap is caught
here instead of on the one 8-GPU attempt: a world=8 tiny dry run, a
world=8 fedavg dry run (leaders + member ranks + intra-party groups), and
asymmetric sub-groups (the 2+3+3 split of BASELINE config 5 needs
communicators that do not span the mesh — SURVEY §7 hard parts).
"""
import json
import multiprocessing
import os
import subprocess
import sys

import pytest

torch = pytest.importorskip("torch")

from tests._util import free_ports  # noqa: E402

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_mp = multiprocessing.get_context("spawn")


def _torchrun_bench(extra_args, world=8, timeout=420, attempts=2):
    # Two attempts with fresh ports: the probed-free lane/master ports can
    # be grabbed by a concurrent ephemeral connection between probe and
    # bind (rare CI flake) — a retry re-rolls every port.
    for _ in range(attempts):
        port = free_ports(1)[0]
        env = dict(os.environ)
        env["RAYFED_BENCH_EXTRAS"] = "0"
        env["RAYFED_BENCH_BASE_PORT"] = str(free_ports(1)[0] + 2000)
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={world}",
            "--master-addr", "127.0.0.1", "--master-port", str(port),
            os.path.join(REPO, "bench.py"), "--gpus", str(world),
        ] + extra_args
        r = subprocess.run(
            cmd, capture_output=True, text=True, timeout=timeout, env=env,
            cwd=REPO,
        )
        if r.returncode == 0:
            break
    assert r.returncode == 0, f"torchrun failed:\n{r.stdout}\n{r.stderr}"
    line = next(
        ln for ln in r.stdout.strip().splitlines() if ln.startswith("{")
    )
    return json.loads(line)


def test_torchrun_world8_tiny_dry_run():
    """The driver's N=8 tiny launch, on CPU: 4 lanes x 2 parties."""
    j = _torchrun_bench(["--steps", "4", "--warmup", "1"])
    assert j["n_gpus"] == 8
    assert j["metric"] == "cross_party_objects_per_sec"
    assert j["value"] > 0
    assert j["transport"] in ("cpp", "asyncio")
    assert "x4 lanes" in j["config"]["parallelism"]


def test_torchrun_world8_fedavg_dry_run():
    """The driver's N=8 fedavg launch, on CPU: 2 leaders + 6 member ranks,
    intra-party all-reduce groups of 4."""
    j = _torchrun_bench([
        "--mode", "fedavg", "--steps", "2", "--warmup", "1",
        "--layers", "0", "--vocab", "4096",
    ])
    assert j["n_gpus"] == 8
    assert j["metric"] == "fedavg_cross_party_GBps"
    assert j["value"] > 0
    assert "x 4 GPUs" in j["config"]["parallelism"]


def _subgroup_worker(rank, world, port, q):
    import torch
    import torch.distributed as dist

    from rayfed_amd.parallel.group import destroy_party_group, init_party_group

    init_party_group(rank, world, master_port=port, backend="gloo")
    # Asymmetric party split: sizes 1 + 3 (shape of the 2+3+3 BASELINE
    # config 5 split — groups that do NOT span the full mesh).  Every rank
    # must execute every new_group call in the same order.
    g_solo = dist.new_group([0], backend="gloo")
    g_rest = dist.new_group([1, 2, 3], backend="gloo")
    # Symmetric pairs too (the N=4 scaling shape).
    g_lo = dist.new_group([0, 1], backend="gloo")
    g_hi = dist.new_group([2, 3], backend="gloo")

    t = torch.full((64,), float(rank + 1))
    mine = g_solo if rank == 0 else g_rest
    dist.all_reduce(t, group=mine)
    expect = 1.0 if rank == 0 else float(2 + 3 + 4)
    ok1 = torch.allclose(t, torch.full((64,), expect))

    t2 = torch.full((8,), float(rank))
    pair = g_lo if rank < 2 else g_hi
    dist.all_reduce(t2, group=pair)
    expect2 = float(0 + 1) if rank < 2 else float(2 + 3)
    ok2 = torch.allclose(t2, torch.full((8,), expect2))
    q.put((rank, ok1 and ok2))
    destroy_party_group()


def test_asymmetric_subgroups_world4():
    port = free_ports(1)[0]
    q = _mp.Queue()
    procs = [
        _mp.Process(target=_subgroup_worker, args=(r, 4, port, q))
        for r in range(4)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, ok = q.get(timeout=180)
        results[rank] = ok
    for p in procs:
        p.join(timeout=30)
    assert results == {0: True, 1: True, 2: True, 3: True}
