#!/usr/bin/env python3
"""Summarize rocprofv3 rocpd .db outputs (kernel-trace / --pmc runs) into
the committed per-kernel text summaries under profiles/.

Usage: python tools/rocpd_summarize.py gpurun_out/prof/ktrace_results.db
       python tools/rocpd_summarize.py gpurun_out/prof/fetch_results.db --pmc
"""
import collections
import re
import sqlite3
import sys


def tables(con):
    out = {}
    for (name,) in con.execute(
            "select name from sqlite_master where type='table'"):
        base = re.sub(r"_[0-9a-f]{8}_.*$", "", name).replace("rocpd_", "")
        out[base] = name
    return out


def short(name):
    name = re.sub(r"^_Z\d+", "", name)
    name = name.split("(")[0].split("IL")[0].split("Ij")[0].split("Im")[0]
    return name[:60]


def kernel_stats(path):
    con = sqlite3.connect(path)
    t = tables(con)
    q = f"""
      select s.kernel_name, d.start, d.end,
             d.grid_size_x, d.workgroup_size_x
      from {t['kernel_dispatch']} d
      join {t['info_kernel_symbol']} s on s.id = d.kernel_id
    """
    agg = collections.defaultdict(lambda: [0, 0.0, 0.0])
    for name, start, end, gx, wx in con.execute(q):
        ms = (end - start) / 1e6
        a = agg[short(name)]
        a[0] += 1
        a[1] += ms
        a[2] = max(a[2], ms)
    total = sum(a[1] for a in agg.values())
    rows = sorted(agg.items(), key=lambda kv: -kv[1][1])
    out = [f"{'kernel':60s} {'calls':>6s} {'total_ms':>10s} "
           f"{'avg_ms':>9s} {'max_ms':>9s} {'pct':>6s}"]
    for name, (n, tot, mx) in rows:
        out.append(f"{name:60s} {n:6d} {tot:10.3f} {tot / n:9.3f} "
                   f"{mx:9.3f} {100 * tot / total:5.1f}%")
    out.append(f"{'TOTAL':60s} {'':6s} {total:10.3f}")
    return "\n".join(out)


def pmc_stats(path):
    con = sqlite3.connect(path)
    t = tables(con)
    pcols = [r[1] for r in con.execute(
        f"PRAGMA table_info({t['pmc_event']})")]
    # pmc_event links event_id -> value; dispatch has event_id
    q = f"""
      select s.kernel_name, i.name, p.value, d.start, d.end
      from {t['pmc_event']} p
      join {t['info_pmc']} i on i.id = p.pmc_id
      join {t['kernel_dispatch']} d on d.event_id = p.event_id
      join {t['info_kernel_symbol']} s on s.id = d.kernel_id
    """
    agg = collections.defaultdict(lambda: collections.defaultdict(
        lambda: [0, 0.0, 0.0]))
    try:
        rows = list(con.execute(q))
    except sqlite3.OperationalError as e:
        return f"pmc query failed: {e}\npmc_event cols: {pcols}"
    for kname, cname, value, start, end in rows:
        a = agg[short(kname)][cname]
        a[0] += 1
        a[1] += value
        a[2] += (end - start) / 1e6
    out = [f"{'kernel':60s} {'counter':>12s} {'calls':>6s} "
           f"{'avg_value':>14s} {'avg_ms':>9s}"]
    for kname, counters in sorted(agg.items()):
        for cname, (n, tot, ms) in counters.items():
            out.append(f"{kname:60s} {cname:>12s} {n:6d} "
                       f"{tot / n:14.1f} {ms / n:9.3f}")
    return "\n".join(out)


if __name__ == "__main__":
    path = sys.argv[1]
    if "--pmc" in sys.argv:
        print(pmc_stats(path))
    else:
        print(kernel_stats(path))
