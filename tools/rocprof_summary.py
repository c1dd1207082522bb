#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db into a per-kernel table.

    python tools/rocprof_summary.py gpurun_out/prof3/alex3_results.db

Used to produce the committed profiles/*.md evidence.
"""

import sqlite3
import sys


def summarize(dbfile, top=30):
    db = sqlite3.connect(dbfile)
    cur = db.cursor()
    sfx = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")][0] \
        .replace("rocpd_kernel_dispatch_", "")
    q = f"""
    SELECT ks.display_name, COUNT(*),
           SUM(k.end-k.start)/1e6, AVG(k.end-k.start)/1e3,
           MAX(k.grid_size_x*k.grid_size_y*k.grid_size_z /
               (k.workgroup_size_x*k.workgroup_size_y*k.workgroup_size_z))
    FROM rocpd_kernel_dispatch_{sfx} k
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {top}
    """
    rows = cur.execute(q).fetchall()
    tot = sum(r[2] for r in rows)
    out = []
    out.append("| kernel | calls | total ms | avg µs | workgroups | % |")
    out.append("|---|---|---|---|---|---|")
    for name, calls, ms, avg, wgs in rows:
        short = name.split("(")[0][:60]
        out.append("| `%s` | %d | %.2f | %.1f | %d | %.1f |" %
                   (short, calls, ms, avg, wgs, 100 * ms / tot))
    out.append("")
    out.append("total kernel time: %.1f ms" % tot)
    return "\n".join(out)


if __name__ == "__main__":
    print(summarize(sys.argv[1],
                    int(sys.argv[2]) if len(sys.argv) > 2 else 30))
