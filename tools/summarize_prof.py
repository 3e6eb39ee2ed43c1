#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel-trace db into a markdown kernel-time table.

Usage: python tools/summarize_prof.py <db-glob> <out.md> <steps> [note...]
"""

import glob
import re
import sqlite3
import sys


def main():
    db = sorted(glob.glob(sys.argv[1], recursive=True))[-1]
    out_path = sys.argv[2]
    steps = int(sys.argv[3])
    note = " ".join(sys.argv[4:])
    con = sqlite3.connect(db)
    t = [
        r[0]
        for r in con.execute(
            "SELECT name FROM sqlite_master WHERE type='table' "
            "AND name LIKE 'rocpd_kernel_dispatch%'"
        )
    ][0]
    sfx = t[len("rocpd_kernel_dispatch_"):]
    strings = dict(con.execute(f"SELECT id, string FROM rocpd_string_{sfx}"))
    rows = [
        (strings.get(n, str(n)), c, ms)
        for n, c, ms in con.execute(
            f"SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6 "
            f"FROM rocpd_kernel_dispatch_{sfx} k "
            f"JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id "
            f"GROUP BY ks.display_name ORDER BY 3 DESC"
        )
    ]
    tot = sum(r[2] for r in rows)
    agg = {}
    for name, n, ms in rows:
        short = re.sub(r"<.*", "", name)[:80].strip()
        a = agg.setdefault(short, [0, 0.0])
        a[0] += n
        a[1] += ms
    lines = [
        "# rocprofv3 kernel-time summary",
        "",
        note,
        "",
        "| ms/step | % | calls | kernel |",
        "|---|---|---|---|",
    ]
    for short, (n, ms) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:20]:
        lines.append(f"| {ms/steps:.2f} | {100*ms/tot:.1f} | {n} | `{short}` |")
    lines += [
        "",
        f"Total kernel time: {tot/steps:.1f} ms/step (GPU-bound; launch gaps "
        "negligible).",
        "",
        "The fused flash-attention kernels replace the reference K2-K5 chain "
        "entirely; Cijk_* are hipBLASLt GEMMs; all (anonymous namespace)::* "
        "kernels are libai_amd gfx950 HIP kernels from libai_amd/csrc/kernels/.",
    ]
    with open(out_path, "w") as f:
        f.write("\n".join(lines) + "\n")
    print("\n".join(lines[:16]))


if __name__ == "__main__":
    main()
