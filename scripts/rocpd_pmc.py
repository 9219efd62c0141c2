#!/usr/bin/env python3
"""Aggregate rocprofv3 --pmc counters per kernel from a rocpd .db.

    python scripts/rocpd_pmc.py gpurun_out/pmc/x_results.db [out.txt]

Prints, per kernel name, the SUM of each counter over all dispatches plus
dispatch count — enough to compute per-dispatch averages and ratios
(e.g. SQ_WAIT_ANY / SQ_WAVE_CYCLES).
"""
import re
import sqlite3
import sys


def main(path, out=None):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [n for (n,) in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    ctab = [t for t in tables
            if "counter" in t.lower() or "pmc_event" in t.lower()]
    dtab = [t for t in tables if "kernel_dispatch" in t]
    if not ctab:
        text = "no counter table; tables:\n" + "\n".join(tables)
        print(text)
        if out:
            open(out, "w").write(text + "\n")
        sys.exit(1)
    lines = []
    for t in ctab:
        cols = [r[1] for r in cur.execute(f"PRAGMA table_info({t})")]
        lines.append(f"-- {t}: {cols}")
    # rocpd layout: counters table has (dispatch) id + counter_id + value;
    # kernel names via the dispatch/info views
    view = [t for t in tables if t.startswith("rocpd_info_kernel_symbol")]
    try:
        uuid = re.match(r"rocpd_kernel_dispatch_(.*)", dtab[0]).group(1)
        q = f"""
        SELECT s.display_name, c.counter_name, SUM(c.value), COUNT(*)
        FROM counters_collection_{uuid} c
        JOIN rocpd_info_kernel_symbol_{uuid} s
          ON s.id = c.kernel_id
        GROUP BY s.display_name, c.counter_name"""
        rows = list(cur.execute(q))
    except Exception as e:
        lines.append(f"primary query failed: {e}")
        rows = []
        # fallback: try the rocpd_pmc_event layout (ROCm 7.x rocpd)
        try:
            q = f"""
            SELECT s.display_name, i.name, SUM(p.value), COUNT(*)
            FROM rocpd_pmc_event_{uuid} p
            JOIN rocpd_info_pmc_{uuid} i ON i.id = p.pmc_id
            JOIN rocpd_event_{uuid} e ON e.id = p.event_id
            JOIN rocpd_kernel_dispatch_{uuid} d ON d.event_id = e.id
            JOIN rocpd_info_kernel_symbol_{uuid} s ON s.id = d.kernel_id
            GROUP BY s.display_name, i.name"""
            rows = list(cur.execute(q))
        except Exception as e2:
            lines.append(f"fallback query failed: {e2}")
            for t in tables:
                if any(k in t for k in ("pmc", "kernel_dispatch", "event",
                                        "kernel_symbol")):
                    cols = [r[1] for r in
                            cur.execute(f"PRAGMA table_info({t})")]
                    lines.append(f"{t}:\n  cols={cols}")
                    for r in cur.execute(f"SELECT * FROM {t} LIMIT 2"):
                        lines.append(f"  row={r!r:.300s}")
    agg = {}
    for name, cname, val, n in rows:
        short = name.replace("(anonymous namespace)::", "")
        short = short.split("(")[0][:70]
        agg.setdefault(short, {})[cname] = (val, n)
    for k, counters in sorted(
            agg.items(),
            key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", (0, 0))[0]):
        lines.append(f"== {k}")
        wc = counters.get("SQ_WAVE_CYCLES", (0, 1))[0] or 1
        for cname, (val, n) in sorted(counters.items()):
            lines.append(
                f"   {cname:28s} sum={val:16.0f} n={n:5d} "
                f"ratio_vs_wavecyc={val / wc:7.4f}")
    text = "\n".join(lines)
    print(text)
    if out:
        open(out, "w").write(text + "\n")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
