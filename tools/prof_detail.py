#!/usr/bin/env python3
"""Detailed rocprofv3 db analysis.

Usage:
  python tools/prof_detail.py kernels <db-glob> <steps>   # name+grid breakdown
  python tools/prof_detail.py pmc <db-glob> <name-filter> # per-kernel counters
"""

import glob
import re
import sqlite3
import sys


def open_db(pat):
    db = sorted(glob.glob(pat, recursive=True))[-1]
    con = sqlite3.connect(db)
    t = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")][0]
    return con, t[len("rocpd_kernel_dispatch_"):]


def kernels(pat, steps):
    con, sfx = open_db(pat)
    strings = dict(con.execute(f"SELECT id, string FROM rocpd_string_{sfx}"))
    rows = con.execute(
        f"SELECT ks.display_name, k.grid_size_x, k.grid_size_y, k.grid_size_z,"
        f" COUNT(*), SUM(k.end-k.start)/1e6 "
        f"FROM rocpd_kernel_dispatch_{sfx} k "
        f"JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id "
        f"GROUP BY ks.display_name, k.grid_size_x, k.grid_size_y, k.grid_size_z "
        f"ORDER BY 6 DESC LIMIT 40").fetchall()
    for n, gx, gy, gz, c, ms in rows:
        name = strings.get(n, str(n))
        name = re.sub(r"<[^>]*>", "", name)[:70]
        print(f"{ms/steps:7.2f} ms/step  x{c:5d}  grid[{gx},{gy},{gz}]  {name}")


def pmc(pat, filt):
    con, sfx = open_db(pat)
    strings = dict(con.execute(f"SELECT id, string FROM rocpd_string_{sfx}"))
    # schema introspection (rocprofv3 writes rocpd_pmc_event + rocpd_info_pmc)
    def cols(t):
        return [r[1] for r in con.execute(f"PRAGMA table_info({t})")]
    ev = f"rocpd_pmc_event_{sfx}"
    info = f"rocpd_info_pmc_{sfx}"
    try:
        evc, infc = cols(ev), cols(info)
        print("# pmc_event cols:", evc, file=sys.stderr)
        print("# info_pmc cols:", infc, file=sys.stderr)
        # best-effort join: event has (pmc_id, value, <dispatch link>)
        dispatch_col = next(c for c in evc if "dispatch" in c or c == "event_id")
        name_col = "name" if "name" in infc else infc[1]
        rows = con.execute(
            f"SELECT ks.display_name, di.{name_col}, SUM(e.value), COUNT(*) "
            f"FROM {ev} e "
            f"JOIN rocpd_kernel_dispatch_{sfx} k ON e.{dispatch_col}=k.id "
            f"JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id "
            f"JOIN {info} di ON e.pmc_id=di.id "
            f"GROUP BY 1, 2").fetchall()
    except (sqlite3.OperationalError, StopIteration) as e:
        print("schema probe:", e)
        for r in con.execute("SELECT name FROM sqlite_master WHERE type='table'"):
            if 'counter' in r[0] or 'pmc' in r[0]:
                t = r[0]
                print("  table:", t, cols(t))
        return
    agg = {}
    for n, cn, v, cnt in rows:
        name = re.sub(r"<[^>]*>", "", strings.get(n, str(n)))[:50]
        if filt.lower() not in name.lower():
            continue
        agg.setdefault(name, {})[cn] = v
    for name, d in agg.items():
        print(name)
        for cn, v in sorted(d.items()):
            print(f"    {cn:28s} {v:,.0f}")
        cyc = d.get("SQ_WAVE_CYCLES")
        for k in ("SQ_WAIT_ANY", "SQ_WAIT_INST_ANY", "SQ_ACTIVE_INST_ANY"):
            if cyc and d.get(k):
                print(f"    {k}/CYCLES = {d[k]/cyc:.2%}")
        if d.get("SQ_INSTS_MFMA") and d.get("SQ_ACTIVE_INST_ANY"):
            print(f"    instr/MFMA = {d['SQ_ACTIVE_INST_ANY']/d['SQ_INSTS_MFMA']:.1f}")


if __name__ == "__main__":
    if sys.argv[1] == "kernels":
        kernels(sys.argv[2], int(sys.argv[3]))
    else:
        pmc(sys.argv[2], sys.argv[3] if len(sys.argv) > 3 else "")
