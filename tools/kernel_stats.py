"""Summarize per-kernel time from a rocprofv3 results.db (kernel-trace).

Usage: python tools/kernel_stats.py gpurun_out/prof/matrix_results.db
Writes a markdown table to stdout (redirect into profiles/)."""

import sqlite3
import sys
from collections import defaultdict


def summarize(path: str) -> str:
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    uid = disp[len("rocpd_kernel_dispatch_"):]
    sym = f"rocpd_info_kernel_symbol_{uid}"
    q = f"""
      SELECT k.display_name, COUNT(*), SUM(d.end - d.start),
             AVG(d.end - d.start)
      FROM {disp} d
      JOIN {sym} k ON d.kernel_id = k.id
      GROUP BY k.display_name ORDER BY SUM(d.end - d.start) DESC
    """
    rows = list(db.execute(q))
    total = sum(r[2] for r in rows) or 1
    out = ["| kernel | calls | total ms | avg us | % |",
           "|---|---|---|---|---|"]
    for name, calls, tot, avg in rows:
        short = name.split("(")[0][:70]
        out.append(f"| {short} | {calls} | {tot/1e6:.3f} | "
                   f"{avg/1e3:.2f} | {100*tot/total:.1f} |")
    return "\n".join(out)


if __name__ == "__main__":
    for p in sys.argv[1:]:
        print(f"## {p}\n")
        print(summarize(p))
        print()
