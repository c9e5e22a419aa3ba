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


def _entry():
    if len(sys.argv) > 2 and sys.argv[2] == "streams":
        return  # handled at module bottom
    main(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 20)


if __name__ == "__main__":
    _entry()


def streams(db):
    """Per-stream busy time + pairwise overlap (second arg 'streams')."""
    con = sqlite3.connect(db)
    names = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [n for n in names if n.startswith("rocpd_kernel_dispatch")][0] \
        .replace("rocpd_kernel_dispatch_", "")
    rows = list(con.execute(
        f"SELECT stream_id, start, end FROM rocpd_kernel_dispatch_{sfx} "
        f"ORDER BY start"))
    t0 = min(r[1] for r in rows)
    t1 = max(r[2] for r in rows)
    per = {}
    for sid, s, e in rows:
        per.setdefault(sid, []).append((s, e))

    def merge(iv):
        out = []
        for s, e in iv:
            if out and s <= out[-1][1]:
                out[-1] = (out[-1][0], max(out[-1][1], e))
            else:
                out.append((s, e))
        return out

    merged = {k: merge(v) for k, v in per.items()}
    print(f"window {((t1-t0)/1e6):.1f} ms")
    for k, iv in sorted(merged.items(),
                        key=lambda kv: -sum(e-s for s, e in kv[1])):
        busy = sum(e - s for s, e in iv) / 1e6
        print(f"stream {k}: busy {busy:.1f} ms ({len(per[k])} dispatches)")
    ks = sorted(merged, key=lambda k: -sum(e-s for s, e in merged[k]))[:3]
    for i in range(len(ks)):
        for j in range(i + 1, len(ks)):
            a, b = merged[ks[i]], merged[ks[j]]
            ov = ai = bi = 0
            x = 0.0
            while ai < len(a) and bi < len(b):
                s = max(a[ai][0], b[bi][0])
                e = min(a[ai][1], b[bi][1])
                if e > s:
                    ov += e - s
                if a[ai][1] < b[bi][1]:
                    ai += 1
                else:
                    bi += 1
            print(f"overlap {ks[i]}x{ks[j]}: {ov/1e6:.1f} ms")


if len(sys.argv) > 2 and sys.argv[2] == "streams":
    streams(sys.argv[1])
