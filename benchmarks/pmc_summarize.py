#!/usr/bin/env python3
"""Aggregate a rocprofv3 PMC .db per kernel and print a compact table
(run on the GPU box so only text comes back, not multi-MB databases)."""
import sqlite3
import sys


def main(db, top=12):
    con = sqlite3.connect(db)
    names = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [n for n in names if n.startswith("rocpd_pmc_event")][0] \
        .replace("rocpd_pmc_event_", "")
    q = f"""
    SELECT ks.display_name, ip.name, SUM(pe.value), COUNT(DISTINCT kd.id)
    FROM rocpd_pmc_event_{sfx} pe
    JOIN rocpd_info_pmc_{sfx} ip ON pe.pmc_id = ip.id
    JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.event_id = kd.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name, ip.name
    """
    per = {}
    for disp, cname, val, ndisp in con.execute(q):
        k = str(disp)[:56]
        per.setdefault(k, {})[cname] = val
        per[k]["_n"] = ndisp
    # rank kernels by wave cycles (or first counter)
    def keyf(kv):
        return -(kv[1].get("SQ_WAVE_CYCLES") or
                 sum(v for c, v in kv[1].items() if c != "_n"))
    for k, cs in sorted(per.items(), key=keyf)[:top]:
        print(f"== {k} (n={cs.pop('_n')})")
        for c, v in sorted(cs.items()):
            print(f"   {c:26} {v:,.0f}")


if __name__ == "__main__":
    main(sys.argv[1])
