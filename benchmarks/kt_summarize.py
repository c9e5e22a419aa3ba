#!/usr/bin/env python3
"""Aggregate a rocprofv3 kernel-trace .db per kernel (run on the GPU box;
only text comes back)."""
import sqlite3
import sys


def main(db, top=20):
    con = sqlite3.connect(db)
    names = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [n for n in names if n.startswith("rocpd_kernel_dispatch")][0] \
        .replace("rocpd_kernel_dispatch_", "")
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e3,
           AVG(kd.end-kd.start)/1e3
    FROM rocpd_kernel_dispatch_{sfx} kd
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {top}
    """
    for r in con.execute(q):
        print(f"{str(r[0])[:70]:70} n={r[1]:5} tot={r[2]:9.1f}us "
              f"avg={r[3]:7.1f}us")


if __name__ == "__main__":
    main(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 20)
