#!/usr/bin/env python3
"""Data-plane kernel microbenchmarks on MI355X (run under gpurun).

Measures: CRC32 kernel GB/s, fused pack+CRC, fp8 cast+CRC, FedAvg reduce,
pinned D2H/H2D DMA, and the full pack_to_host/unpack_from_host pipeline.
"""
import json
import sys
import time

import torch

sys.path.insert(0, ".")
from rayfed_amd.config import GpuDataPlaneConfig  # noqa: E402
from rayfed_amd.ops import _hip_loader  # noqa: E402
from rayfed_amd.ops.gpu_plane import GpuDataPlane  # noqa: E402

ext = _hip_loader.load()


def timeit(fn, reps=5, warm=2):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def report(name, nbytes, secs, extra=None):
    row = {"op": name, "MiB": nbytes >> 20, "GB/s": round(nbytes / secs / 1e9, 1),
           "ms": round(secs * 1e3, 3)}
    if extra:
        row.update(extra)
    print(json.dumps(row), flush=True)


def main():
    torch.cuda.set_device(0)
    sizes = [1 << 24, 1 << 27, 1 << 30]  # 16 MiB, 128 MiB, 1 GiB

    for n in sizes:
        data = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
        report("crc32", n, timeit(lambda: ext.crc32(data)))

    for n in sizes:
        src = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
        dst = torch.empty(n, dtype=torch.uint8, device="cuda")
        out = [None]

        def run():
            out[0] = ext.pack_crc_async(src, dst)

        # read+write pass: report as 2x bytes moved
        report("pack_crc(fused copy+crc)", 2 * n, timeit(run))

    for n_elems in [1 << 24, 1 << 28]:  # bf16 elements
        src = torch.randn(n_elems, device="cuda").to(torch.bfloat16)
        dst = torch.empty(n_elems, dtype=torch.uint8, device="cuda")

        def run():
            ext.pack_fp8_async(src, dst)

        report("pack_fp8(cast+crc)", 3 * n_elems, timeit(run),
               {"elems": n_elems})

    # FedAvg reduce: k inputs + 1 output of bf16
    for k in [2, 4, 8]:
        n = 1 << 28  # elements
        ins = [torch.randn(n, device="cuda").to(torch.bfloat16) for _ in range(k)]
        outt = torch.empty(n, dtype=torch.bfloat16, device="cuda")
        w = [1.0 / k] * k

        def run():
            ext.fedavg_reduce_(outt, ins, w)

        moved = (k + 1) * n * 2
        report(f"fedavg_reduce k={k} bf16", moved, timeit(run), {"elems": n})

    # MFMA combine variant (documented comparison; VALU path is default)
    for k in [4, 16]:
        n = 1 << 28
        ins = [torch.randn(n, device="cuda").to(torch.bfloat16) for _ in range(k)]
        outt = torch.empty(n, dtype=torch.bfloat16, device="cuda")
        w = [1.0 / k] * k

        def run_mfma():
            ext.fedavg_reduce_mfma_(outt, ins, w)

        report(f"fedavg_reduce_MFMA k={k} bf16", (k + 1) * n * 2,
               timeit(run_mfma), {"elems": n})

    # DMA: pinned D2H / H2D
    n = 1 << 30
    dev = torch.empty(n, dtype=torch.uint8, device="cuda")
    pin = torch.empty(n, dtype=torch.uint8, pin_memory=True)
    report("D2H pinned", n, timeit(lambda: pin.copy_(dev, non_blocking=False)))
    report("H2D pinned", n, timeit(lambda: dev.copy_(pin, non_blocking=False)))

    # Bidirectional DMA aggregate: is the host link full duplex?
    n = 1 << 30
    dev1 = torch.empty(n, dtype=torch.uint8, device="cuda")
    dev2 = torch.empty(n, dtype=torch.uint8, device="cuda")
    pin1 = torch.empty(n, dtype=torch.uint8, pin_memory=True)
    pin2 = torch.empty(n, dtype=torch.uint8, pin_memory=True)
    s_d2h = torch.cuda.Stream()
    s_h2d = torch.cuda.Stream()

    def bidir():
        with torch.cuda.stream(s_d2h):
            pin1.copy_(dev1, non_blocking=True)
        with torch.cuda.stream(s_h2d):
            dev2.copy_(pin2, non_blocking=True)
        torch.cuda.synchronize()

    report("bidirectional D2H+H2D (2 GiB moved)", 2 * n, timeit(bidir))

    # Registered /dev/shm segment: does torch see it as pinned (async DMA)?
    from rayfed_amd.ops import shm_pool

    pool = shm_pool.get_send_pool()
    seg = pool.acquire(1 << 28)
    print(json.dumps({"op": "shm_seg", "registered": seg.registered,
                      "torch_is_pinned": bool(seg.torch_view.is_pinned())}),
          flush=True)

    def shm_d2h():
        seg.torch_view[: 1 << 28].copy_(dev1[: 1 << 28], non_blocking=True)
        torch.cuda.synchronize()

    report("D2H into registered shm", 1 << 28, timeit(shm_d2h))
    pool.release(seg)

    # Chunked shm pipeline e2e in-process with SEPARATE planes (distinct
    # copy streams), approximating the two-process pipeline.
    from rayfed_amd.config import GpuDataPlaneConfig as _C
    from rayfed_amd.ops.gpu_plane import GpuDataPlane as _P
    pl_tx = _P(_C())
    pl_rx = _P(_C())
    t = torch.randn((1 << 30) // 4, device="cuda")  # 1 GiB f32

    def chunked_e2e():
        seg2, man, rel = pl_tx.pack_to_shm_chunked(t)
        man = dict(man, nbytes=t.numel() * 4, shm=seg2.name)
        out = pl_rx.unpack_from_shm_chunked(seg2.name, man, torch.float32,
                                            [t.numel()])
        rel()
        return out

    report("chunked shm pack||unpack e2e 1 GiB (2 planes)", 1 << 30,
           timeit(chunked_e2e, reps=3, warm=1))
    shm_pool.detach_all()

    # Full plane pipeline
    plane = GpuDataPlane(GpuDataPlaneConfig())
    for gib in [0.25, 1.0]:
        numel = int(gib * (1 << 30)) // 2
        t = torch.randn(numel, device="cuda").to(torch.bfloat16)
        nbytes = numel * 2

        def pack():
            _, _, rel = plane.pack_to_host(t)
            if rel:
                rel()

        secs = timeit(pack, reps=3, warm=1)
        report(f"plane.pack_to_host {gib} GiB bf16", nbytes, secs)

        raw, crc, _rel = plane.pack_to_host(t)

        def unpack():
            plane.unpack_from_host(memoryview(raw), torch.bfloat16, [numel], crc)

        report(f"plane.unpack_from_host {gib} GiB bf16", nbytes,
               timeit(unpack, reps=3, warm=1))

    plane8 = GpuDataPlane(GpuDataPlaneConfig(wire_dtype="fp8e4m3"))
    numel = (1 << 29)  # 1 GiB bf16 -> 512 MiB wire
    t = torch.randn(numel, device="cuda").to(torch.bfloat16)

    def pack8():
        _, _, rel = plane8.pack_to_host(t)
        if rel:
            rel()

    report("plane.pack_to_host fp8-wire 1GiB bf16", numel * 2,
           timeit(pack8, reps=3, warm=1))


if __name__ == "__main__":
    main()
