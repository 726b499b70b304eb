#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite database into per-kernel stats.

Usage: python scripts/rocpd_stats.py <dir-or-db> [top_n]

rocprofv3 --kernel-trace --stats -d DIR writes one .db per traced
process (rocpd schema, table names suffixed with a per-run uid).  This
tool aggregates kernel dispatch time per kernel name across every db it
finds and prints a table sorted by total time — the shape committed
under profiles/ for the judge (see BASELINE.md "profiling recipe").

Schema handling is introspective (column names vary slightly across
ROCm releases): the dispatch table's timestamp columns are matched by
name, the kernel name comes from the info_kernel_symbol table joined on
kernel id.
"""
from __future__ import annotations

import glob
import os
import sqlite3
import sys
from collections import defaultdict


def _cols(con, table):
    return [r[1] for r in con.execute(f"PRAGMA table_info({table})")]


def _pick(cols, *cands):
    for c in cands:
        for col in cols:
            if c in col.lower():
                return col
    return None


def stats_from_db(path, acc):
    con = sqlite3.connect(f"file:{path}?mode=ro", uri=True)
    try:
        tables = [
            r[0]
            for r in con.execute(
                "SELECT name FROM sqlite_master WHERE type='table'"
            )
        ]
        for disp in [t for t in tables if t.startswith("rocpd_kernel_dispatch")]:
            uid = disp[len("rocpd_kernel_dispatch"):]
            sym = "rocpd_info_kernel_symbol" + uid
            if sym not in tables:
                continue
            dcols = _cols(con, disp)
            scols = _cols(con, sym)
            start = _pick(dcols, "start")
            end = _pick(dcols, "end")
            kid = _pick(dcols, "kernel_id", "symbol")
            sid = _pick(scols, "id")
            name = _pick(scols, "formatted_kernel_name", "kernel_name", "name")
            if not all((start, end, kid, sid, name)):
                continue
            q = (
                f"SELECT s.{name}, COUNT(*), SUM(d.{end}-d.{start}) "
                f"FROM {disp} d JOIN {sym} s ON d.{kid}=s.{sid} "
                f"GROUP BY s.{name}"
            )
            for kname, calls, total_ns in con.execute(q):
                acc[kname][0] += calls
                acc[kname][1] += total_ns or 0
    finally:
        con.close()


def main():
    target = sys.argv[1] if len(sys.argv) > 1 else "."
    top_n = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    dbs = (
        [target]
        if target.endswith(".db")
        else sorted(glob.glob(os.path.join(target, "**", "*.db"), recursive=True))
    )
    if not dbs:
        print(f"no rocpd .db files under {target}", file=sys.stderr)
        return 1
    acc = defaultdict(lambda: [0, 0])
    for db in dbs:
        stats_from_db(db, acc)
    rows = sorted(acc.items(), key=lambda kv: -kv[1][1])[:top_n]
    print(f"{'kernel':<64} {'calls':>7} {'total_ms':>10} {'avg_us':>8}")
    for kname, (calls, total_ns) in rows:
        total_ms = total_ns / 1e6
        avg_us = total_ns / 1e3 / max(1, calls)
        print(f"{str(kname)[:64]:<64} {calls:>7} {total_ms:>10.1f} {avg_us:>8.0f}")
    print(f"\n({len(dbs)} db file(s) aggregated)")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
