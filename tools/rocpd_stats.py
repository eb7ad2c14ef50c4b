#!/usr/bin/env python3
"""Summarise a rocprofv3 rocpd SQLite database into per-kernel stats
(calls, total/avg us, % of GPU time) — the kernel-trace evidence committed
under profiles/."""
import sqlite3
import sys


def kernel_stats(path, top=60):
    db = sqlite3.connect(path)
    suffix = None
    for (name,) in db.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        if name.startswith("rocpd_kernel_dispatch"):
            suffix = name[len("rocpd_kernel_dispatch"):]
    assert suffix, "no kernel dispatch table"
    q = f"""
      SELECT ks.display_name AS kernel, COUNT(*) AS calls,
             SUM(k.end - k.start) AS total_ns,
             AVG(k.end - k.start) AS avg_ns
      FROM rocpd_kernel_dispatch{suffix} k
      JOIN rocpd_info_kernel_symbol{suffix} ks
           ON ks.id = k.kernel_id
      GROUP BY ks.display_name ORDER BY total_ns DESC"""
    try:
        rows = db.execute(q).fetchall()
    except sqlite3.OperationalError:
        cols = [r[1] for r in db.execute(
            f"PRAGMA table_info(rocpd_info_kernel_symbol{suffix})")]
        raise SystemExit(f"schema mismatch; kernel_symbol cols: {cols}")
    total = sum(r[2] for r in rows) or 1
    out = [f"{'%GPU':>6} {'calls':>7} {'total_ms':>10} {'avg_us':>9}  kernel"]
    for name, calls, tot, avg in rows[:top]:
        out.append(f"{100*tot/total:6.2f} {calls:7d} {tot/1e6:10.3f} "
                   f"{avg/1e3:9.2f}  {name[:140]}")
    out.append(f"total GPU kernel time: {total/1e6:.3f} ms over "
               f"{sum(r[1] for r in rows)} dispatches, {len(rows)} kernels")
    return "\n".join(out)


if __name__ == "__main__":
    print(kernel_stats(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 60))
