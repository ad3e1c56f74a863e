"""Summarize a rocprofv3 SQLite result DB into a small markdown table.

The GPU boxes only merge back <= 64 MiB of gpurun_out, and raw trace DBs blow
that budget — so trace/PMC runs post-process on the box:

    rocprofv3 --kernel-trace -d out/prof -- python bench.py ...
    python tools/summarize_trace.py out/prof --steps 4 -o out/trace.md
    rm -rf out/prof

For PMC databases, per-kernel counter sums are reported alongside durations.
"""
from __future__ import annotations

import argparse
import glob
import os
import sqlite3
import sys


def find_db(root: str) -> str:
    cands = sorted(glob.glob(os.path.join(root, "**", "*_results.db"), recursive=True))
    if not cands:
        raise SystemExit(f"no results.db under {root}")
    return cands[0]


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("prof_dir")
    ap.add_argument("--steps", type=int, default=1, help="profiled bench steps (for ms/step)")
    ap.add_argument("--top", type=int, default=30)
    ap.add_argument("-o", "--out", default=None)
    args = ap.parse_args()

    db_path = find_db(args.prof_dir)
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    uid = kd[len("rocpd_kernel_dispatch_"):]

    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(kd.end - kd.start) / 1e6
        FROM rocpd_kernel_dispatch_{uid} kd
        JOIN rocpd_info_kernel_symbol_{uid} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {args.top}
    """).fetchall()
    total = cur.execute(
        f"SELECT SUM(end - start) / 1e6 FROM rocpd_kernel_dispatch_{uid}").fetchone()[0]

    lines = [f"Total kernel time: {total:.1f} ms over {args.steps} profiled steps "
             f"({total / args.steps:.2f} ms/step)", "",
             "| ms/step | calls/step | kernel |", "|---|---|---|"]
    for name, n, ms in rows:
        lines.append(f"| {ms / args.steps:.3f} | {n / args.steps:.0f} | {name[:110]} |")

    # PMC counters, when present
    pmc_tables = [t for t in tables if t.startswith("rocpd_pmc_event")]
    if pmc_tables:
        pe = pmc_tables[0]
        n_ev = cur.execute(f"SELECT COUNT(*) FROM {pe}").fetchone()[0]
        if n_ev:
            cols = [c[1] for c in cur.execute(f"PRAGMA table_info({pe})")]
            lines += ["", f"PMC events: {n_ev} rows, columns {cols}"]
            try:
                pmc_rows = cur.execute(f"""
                    SELECT ks.display_name, pi.name, SUM(pe.value)
                    FROM {pe} pe
                    JOIN rocpd_info_pmc_{uid} pi ON pe.pmc_id = pi.id
                    JOIN rocpd_kernel_dispatch_{uid} kd ON pe.event_id = kd.event_id
                    JOIN rocpd_info_kernel_symbol_{uid} ks ON kd.kernel_id = ks.id
                    GROUP BY ks.display_name, pi.name
                    ORDER BY ks.display_name
                """).fetchall()
                lines += ["", "| kernel | counter | sum |", "|---|---|---|"]
                for k, c, v in pmc_rows[:400]:
                    lines.append(f"| {k[:70]} | {c} | {v:.3e} |")
            except sqlite3.Error as e:
                lines.append(f"(per-kernel PMC join failed: {e})")

    text = "\n".join(lines) + "\n"
    if args.out:
        with open(args.out, "w") as f:
            f.write(text)
    else:
        sys.stdout.write(text)
    return 0


if __name__ == "__main__":
    sys.exit(main())
