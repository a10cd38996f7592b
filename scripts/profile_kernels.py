#!/usr/bin/env python3
"""rocprofv3 capture + summary tooling (SURVEY.md §5 tracing obligation).

Two subcommands:
  kernels  — summarize a rocprofv3 --kernel-trace --stats result db:
             per-kernel call counts / total / avg time / % share
  pmc      — summarize a rocprofv3 --pmc result db: per-kernel counter
             aggregates (per-dispatch averages + derived ratios)

Usage:
  cd /tmp && export TMPDIR=/tmp     # rocprofv3 requirement on these boxes
  rocprofv3 --kernel-trace --stats -d OUT -- python bench.py --steps 8
  python scripts/profile_kernels.py kernels OUT/**/*_results.db
  rocprofv3 --pmc SQ_INSTS_MFMA,SQ_LDS_BANK_CONFLICT,SQ_WAVE_CYCLES,SQ_WAVES \
      --kernel-trace -d OUT2 -- python <workload>
  python scripts/profile_kernels.py pmc OUT2/**/*_results.db --filter attn
NOTE: never combine --pmc with -s/-r/hip/hsa trace domains (pool rule).
"""

from __future__ import annotations

import argparse
import glob
import sqlite3
import sys
from collections import defaultdict


def _open(path_glob: str):
    paths = glob.glob(path_glob, recursive=True) or [path_glob]
    c = sqlite3.connect(paths[0])
    tabs = [r[0] for r in c.execute("select name from sqlite_master where type='table'")]
    disp = [t for t in tabs if t.startswith("rocpd_kernel_dispatch")]
    if not disp:
        sys.exit(f"no kernel dispatch table in {paths[0]}")
    return c, disp[0].replace("rocpd_kernel_dispatch_", "")


def cmd_kernels(args) -> None:
    c, sfx = _open(args.db)
    q = f"""
    SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6, AVG(d.end-d.start)/1e3
    FROM rocpd_kernel_dispatch_{sfx} d
    JOIN rocpd_info_kernel_symbol_{sfx} s ON d.kernel_id = s.id
    GROUP BY s.display_name ORDER BY 3 DESC LIMIT {args.top}
    """
    rows = list(c.execute(q))
    tot = sum(r[2] for r in rows)
    print(f"{'kernel':<70} {'calls':>6} {'ms':>9} {'avg_us':>8} {'%':>5}")
    for name, n, ms, avg in rows:
        print(f"{name[:70]:<70} {n:>6} {ms:>9.2f} {avg:>8.1f} {100 * ms / tot:>5.1f}")
    print(f"total (top {args.top}): {tot:.1f} ms")


def cmd_pmc(args) -> None:
    c, sfx = _open(args.db)
    filt = f"AND s.display_name LIKE '%{args.filter}%'" if args.filter else ""
    q = f"""
    SELECT s.display_name, p.name, SUM(e.value), COUNT(DISTINCT d.id)
    FROM rocpd_pmc_event_{sfx} e
    JOIN rocpd_info_pmc_{sfx} p ON e.pmc_id = p.id
    JOIN rocpd_kernel_dispatch_{sfx} d ON e.event_id = d.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} s ON d.kernel_id = s.id
    WHERE 1=1 {filt}
    GROUP BY s.display_name, p.name
    """
    agg: dict = defaultdict(dict)
    for name, pname, val, n in c.execute(q):
        agg[name[:70]][pname] = (val, n)
    for kname, d in agg.items():
        n = next(iter(d.values()))[1]
        print(f"\n{kname}  ({n} dispatches)")
        vals = {}
        for pname, (val, _) in sorted(d.items()):
            vals[pname] = val / n
            print(f"  {pname:<30} {val / n:>18,.0f} /dispatch")
        w = vals.get("SQ_WAVES", 0)
        if w:
            for k, v in vals.items():
                if k != "SQ_WAVES":
                    print(f"  {k + '/wave':<30} {v / w:>18,.1f}")


def main() -> None:
    ap = argparse.ArgumentParser()
    sub = ap.add_subparsers(dest="cmd", required=True)
    k = sub.add_parser("kernels")
    k.add_argument("db")
    k.add_argument("--top", type=int, default=20)
    k.set_defaults(fn=cmd_kernels)
    p = sub.add_parser("pmc")
    p.add_argument("db")
    p.add_argument("--filter", default="")
    p.set_defaults(fn=cmd_pmc)
    args = ap.parse_args()
    args.fn(args)


if __name__ == "__main__":
    main()
