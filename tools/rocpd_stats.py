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




def pmc_stats(path, top=30):
    """Per-kernel PMC aggregates from a rocprofv3 --pmc run: MFMA-busy and
    wait fractions of wave cycles."""
    import collections
    db = sqlite3.connect(path)
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    suffix = [t for t in tabs if t.startswith("rocpd_kernel_dispatch")][0][
        len("rocpd_kernel_dispatch"):]
    pmc_names = dict(db.execute(f"SELECT id, name FROM rocpd_info_pmc{suffix}"))
    q = f"""
      SELECT ks.display_name, p.pmc_id, SUM(p.value), COUNT(*)
      FROM rocpd_pmc_event{suffix} p
      JOIN rocpd_kernel_dispatch{suffix} k ON k.event_id = p.event_id
      JOIN rocpd_info_kernel_symbol{suffix} ks ON ks.id = k.kernel_id
      GROUP BY ks.display_name, p.pmc_id"""
    agg = collections.defaultdict(dict)
    for name, pid, val, cnt in db.execute(q):
        agg[name][pmc_names[pid]] = val
    # report EVERY captured counter (earlier versions hard-coded two and
    # silently dropped e.g. SQ_LDS_BANK_CONFLICT from the output)
    counters = sorted({c for d in agg.values() for c in d})
    ordered = [c for c in ("SQ_WAVE_CYCLES",) if c in counters] + \
        [c for c in counters if c != "SQ_WAVE_CYCLES"]
    rows = sorted(agg.items(), key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", 0))
    hdr = " ".join(f"{c[:14]:>15}" for c in ordered)
    out = [f"{hdr}  kernel",
           "(raw counter sums per kernel; SQ_* are quad-cycles except "
           "SQ_VALU_MFMA_BUSY_CYCLES which counts cycles; derive ratios "
           "against SQ_WAVE_CYCLES)"]
    for name, d in rows[:top]:
        vals = " ".join(f"{d.get(c, 0):15.3e}" for c in ordered)
        out.append(f"{vals}  {name[:110]}")
    return "\n".join(out)


if __name__ == "__main__":
    if len(sys.argv) > 2 and sys.argv[2] == "pmc":
        print(pmc_stats(sys.argv[1]))
    else:
        print(kernel_stats(sys.argv[1],
                           int(sys.argv[2]) if len(sys.argv) > 2 else 60))
