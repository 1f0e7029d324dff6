#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd results.db into a kernel-stats table.

rocprofv3 7.x writes an SQLite "rocpd" database per process; this renders
the per-kernel totals (calls, total/avg us, %) as markdown — the format
committed under profiles/.

    python scripts/rocpd_stats.py gpurun_out/prof_x/runc/729_results.db [N]
"""

import sqlite3
import sys


def stats(db_path: str, top: int = 30):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "select name from sqlite_master where type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = list(cur.execute(f"""
        select s.display_name, count(*), sum(d.end - d.start)
        from {disp} d join {sym} s on d.kernel_id = s.id
        group by s.display_name order by 3 desc"""))
    total = sum(r[2] for r in rows) or 1
    out = ["| kernel | calls | total us | avg us | % |",
           "|---|---|---|---|---|"]
    for name, calls, ns in rows[:top]:
        if len(name) > 72:
            name = name[:69] + "..."
        out.append(f"| `{name}` | {calls} | {ns / 1e3:.0f} "
                   f"| {ns / 1e3 / calls:.1f} | {100 * ns / total:.1f} |")
    out.append(f"\nTotal kernel time {total / 1e6:.2f} ms, "
               f"{sum(r[1] for r in rows)} dispatches, "
               f"{len(rows)} distinct kernels.")
    return "\n".join(out)


if __name__ == "__main__":
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    print(stats(sys.argv[1], top))
