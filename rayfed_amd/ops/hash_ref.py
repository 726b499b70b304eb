"""CPU reference of the device hash64 kernel (csrc/rayfed_hip.hip).

The GPU kernel is FNV-1a over SLOTS = 4 * 524288 interleaved u64 word
streams (slot s owns words {j*SLOTS + s}), a murmur fmix64 finalizer per
lane, an XOR combine across lanes, plus tail-bytes and length terms.  The
mapping depends only on nbytes, so this numpy implementation must produce
bit-identical values — pinned by tests/test_gpu_plane.py on hardware.
Error-detection only; not cryptographic.
"""
from __future__ import annotations

import numpy as np

LANES = 524288  # must equal kHashLanes in csrc/rayfed_hip.hip
SLOTS = 4 * LANES
FNV_OFF = np.uint64(0xCBF29CE484222325)
FNV_P = np.uint64(0x100000001B3)

_M1 = np.uint64(0xFF51AFD7ED558CCD)
_M2 = np.uint64(0xC4CEB9FE1A85EC53)
_S33 = np.uint64(33)


def _fmix64(h):
    h = h ^ (h >> _S33)
    h = h * _M1
    h = h ^ (h >> _S33)
    h = h * _M2
    h = h ^ (h >> _S33)
    return h


def hash64_ref(data: bytes) -> int:
    """64-bit integrity hash of ``data`` — matches hash64_async exactly."""
    old = np.seterr(over="ignore")
    try:
        n = len(data)
        n_words = n // 8
        words = np.frombuffer(data, dtype="<u8", count=n_words)
        # Slot state: h_s = (OFF ^ s) * P, then FNV-1a over the slot's words.
        h = (FNV_OFF ^ np.arange(SLOTS, dtype=np.uint64)) * FNV_P
        for j in range(0, n_words, SLOTS):
            seg = words[j : j + SLOTS]
            k = len(seg)
            h[:k] = (h[:k] ^ seg) * FNV_P
        # Per-lane combine of its 4 slots, then finalize and XOR-reduce.
        hs = h.reshape(LANES, 4)
        hl = (
            ((hs[:, 0] * FNV_P ^ hs[:, 1]) * FNV_P ^ hs[:, 2]) * FNV_P
            ^ hs[:, 3]
        ) * FNV_P
        out = np.bitwise_xor.reduce(_fmix64(hl))
        tail_len = n % 8
        if tail_len:
            tw = np.uint64(0)
            for i, b in enumerate(data[n_words * 8 :]):
                tw |= np.uint64(b) << np.uint64(8 * i)
            out ^= _fmix64((FNV_OFF ^ tw) * FNV_P)
        out ^= _fmix64((np.uint64(n) * FNV_P) ^ FNV_OFF)
        return int(out)
    finally:
        np.seterr(**old)
