"""Summarize a rocprofv3 kernel-trace database into a per-kernel table.

The observability counterpart of the reference's Bösen stats YAML
(SURVEY.md 5.1): after

    rocprofv3 --kernel-trace -d out -- python bench.py ...

run

    python -m poseidon_amd.tools.profile_summary out/**/*_results.db \
        [--iters 13] [--top 20] [--markdown]

to get total/average time and launch counts per kernel, optionally
amortized per iteration.
"""

from __future__ import annotations

import argparse
import glob
import re
import sqlite3


def summarize(db_path: str, iters: int = 1, top: int = 20):
    con = sqlite3.connect(db_path)
    tabs = [r[0] for r in con.execute(
        "select name from sqlite_master where type='table'")]
    kd = [t for t in tabs if t.startswith("rocpd_kernel_dispatch")]
    ks = [t for t in tabs if t.startswith("rocpd_info_kernel_symbol")]
    if not kd or not ks:
        raise SystemExit(f"{db_path}: no rocpd kernel tables "
                         "(did the run use --kernel-trace?)")
    rows = con.execute(
        f"""select s.display_name, count(*),
                   sum(d.end - d.start) / 1e6,
                   avg(d.end - d.start) / 1e3
            from {kd[0]} d join {ks[0]} s on d.kernel_id = s.id
            group by 1 order by 3 desc""").fetchall()
    total_ms = sum(r[2] for r in rows)
    return rows[:top], total_ms


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("db", help="rocprofv3 *_results.db (glob ok)")
    ap.add_argument("--iters", type=int, default=1,
                    help="amortize totals over N iterations")
    ap.add_argument("--top", type=int, default=20)
    ap.add_argument("--markdown", action="store_true")
    args = ap.parse_args(argv)

    paths = sorted(glob.glob(args.db)) or [args.db]
    for path in paths:
        rows, total = summarize(path, args.iters, args.top)
        it = max(args.iters, 1)
        print(f"# {path}: GPU busy {total / it:.3f} ms/iter "
              f"({total:.1f} ms total)")
        if args.markdown:
            print("| ms/iter | % | n/iter | avg us | kernel |")
            print("|---|---|---|---|---|")
        for name, n, ms, us in rows:
            nm = re.sub(r"<[^>]*>", "", name)[:80]
            if args.markdown:
                print(f"| {ms / it:.3f} | {100 * ms / total:.1f} "
                      f"| {n / it:.1f} | {us:.1f} | `{nm}` |")
            else:
                print(f"  {ms / it:8.3f} ms {100 * ms / total:5.1f}%  "
                      f"n={n / it:7.1f}  avg={us:8.1f}us  {nm}")


if __name__ == "__main__":
    main()
