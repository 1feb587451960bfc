#!/usr/bin/env python3
"""Generate a committed profile summary from a rocprofv3 results .db.

Usage: python profiles/make_summary.py <results.db> <tag> "<note>" [steps]
"""

import re
import sqlite3
import sys


def main():
    db, tag, note = sys.argv[1:4]
    steps = int(sys.argv[4]) if len(sys.argv) > 4 else 20
    con = sqlite3.connect(db)
    suf = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE "
        "'rocpd_kernel_dispatch%'")][0].replace("rocpd_kernel_dispatch_", "")
    q = (f"SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 ms, "
         f"AVG(k.end-k.start)/1e3 avg FROM rocpd_kernel_dispatch_{suf} k "
         f"JOIN rocpd_info_kernel_symbol_{suf} ks ON k.kernel_id=ks.id "
         f"GROUP BY 1 ORDER BY ms DESC")
    rows = list(con.execute(q))
    tot = sum(r[2] for r in rows)
    lines = [
        f"# rocprofv3 kernel stats — {tag}",
        f"# {note}",
        f"# flagship bench (CIFAR-10 quirk-ResNet EventGraD, batch 256, "
        f"1x MI355X), {steps} total steps",
        f"# total kernel time: {tot:.1f} ms => {tot/steps:.2f} ms/step",
        f"{'total_ms':>10} {'calls':>6} {'avg_us':>8}  kernel",
    ]
    for name, n, ms, avg in rows[:25]:
        name = re.sub(r"\(.*", "", name)
        lines.append(f"{ms:10.2f} {n:6d} {avg:8.1f}  {name[:80]}")
    out = f"profiles/{tag}.txt"
    open(out, "w").write("\n".join(lines) + "\n")
    print("wrote", out)


if __name__ == "__main__":
    main()
