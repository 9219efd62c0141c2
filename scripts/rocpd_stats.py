#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd .db: total time per kernel, sorted.

    python scripts/rocpd_stats.py gpurun_out/prof/r50b256_results.db [out.md]
"""
import re
import sqlite3
import sys


def summarize(path, out=None):
    db = sqlite3.connect(path)
    cur = db.cursor()
    uuid = None
    for (name,) in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        m = re.match(r"rocpd_kernel_dispatch_(.*)", name)
        if m:
            uuid = m.group(1)
    assert uuid, "no kernel dispatch table"

    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(k.end-k.start),
           AVG(k.end-k.start)
    FROM rocpd_kernel_dispatch_{uuid} k
    JOIN rocpd_info_kernel_symbol_{uuid} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC
    """
    rows = list(cur.execute(q))

    total = sum(r[2] for r in rows) or 1
    lines = [f"{'%':>6s} {'total_ms':>10s} {'calls':>7s} {'avg_us':>9s}  kernel",
             "-" * 100]
    for name, calls, tot, avg in rows[:60]:
        short = re.sub(r"_Z\w*?N12_GLOBAL__N_1", "", str(name))[:78]
        lines.append(f"{100*tot/total:6.2f} {tot/1e6:10.3f} {calls:7d} "
                     f"{avg/1e3:9.1f}  {short}")
    lines.append(f"\nTOTAL GPU kernel time: {total/1e6:.3f} ms "
                 f"over {sum(r[1] for r in rows)} dispatches")
    text = "\n".join(lines)
    print(text)
    if out:
        with open(out, "w") as f:
            f.write(text + "\n")


if __name__ == "__main__":
    summarize(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
