#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite results DB into per-kernel stats.

rocprofv3 on this ROCm 7.2 image writes `<pid>_results.db` (rocpd
format) rather than CSV; this extracts the kernel-dispatch table into
the classic stats view (calls, total/mean/min/max duration).

Usage: python tools/summarize_rocpd.py gpurun_out/prof/runc/*.db > profiles/xxx.md
"""

from __future__ import annotations

import sqlite3
import sys


def table(con: sqlite3.Connection, prefix: str) -> str:
    row = con.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE ?",
        (prefix + "%",),
    ).fetchone()
    if not row:
        raise SystemExit(f"no table {prefix}* in db")
    return row[0]


def summarize(path: str) -> None:
    con = sqlite3.connect(path)
    disp = table(con, "rocpd_kernel_dispatch")
    sym = table(con, "rocpd_info_kernel_symbol")
    strings = table(con, "rocpd_string")

    q = f"""
    SELECT s.display_name, COUNT(*) AS calls,
           SUM(d.end - d.start) / 1000.0 AS total_us,
           AVG(d.end - d.start) / 1000.0 AS mean_us,
           MIN(d.end - d.start) / 1000.0 AS min_us,
           MAX(d.end - d.start) / 1000.0 AS max_us,
           MAX(d.grid_size_x) AS grid_x,
           MAX(d.workgroup_size_x) AS wg_x
    FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
    GROUP BY s.display_name ORDER BY total_us DESC
    """
    try:
        rows = con.execute(q).fetchall()
    except sqlite3.OperationalError:
        # older schema: kernel symbol name via string table
        q = q.replace("s.display_name", "(SELECT string FROM %s WHERE id = s.kernel_name_id)" % strings)
        rows = con.execute(q).fetchall()

    print(f"### {path}")
    print()
    print("| kernel | calls | total µs | mean µs | min µs | max µs | grid.x | wg.x |")
    print("|---|---|---|---|---|---|---|---|")
    for name, calls, total, mean, mn, mx, gx, wx in rows:
        short = (name or "?").split("(")[0][:80]
        print(
            f"| `{short}` | {calls} | {total:.1f} | {mean:.1f} | {mn:.1f} "
            f"| {mx:.1f} | {gx} | {wx} |"
        )
    print()

    # memory copies (present with --memory-copy-trace / sys-trace runs)
    try:
        mc = table(con, "rocpd_memory_copy")
        mc_rows = con.execute(
            f"""SELECT name_id, COUNT(*), SUM(size)/1e9,
                       SUM(end-start)/1e6 FROM {mc} GROUP BY name_id"""
        ).fetchall()
        if mc_rows:
            strt = table(con, "rocpd_string")
            print("| memory copy | count | GB | ms | GB/s |")
            print("|---|---|---|---|---|")
            for name_id, cnt, gb, ms in mc_rows:
                row = con.execute(
                    f"SELECT string FROM {strt} WHERE id=?", (name_id,)
                ).fetchone()
                name = row[0] if row else str(name_id)
                rate = gb / (ms / 1e3) if ms else 0.0
                print(f"| {name} | {cnt} | {gb:.2f} | {ms:.1f} | {rate:.1f} |")
            print()
    except SystemExit:
        pass


if __name__ == "__main__":
    for p in sys.argv[1:]:
        summarize(p)
