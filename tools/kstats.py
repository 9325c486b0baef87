#!/usr/bin/env python3
"""Per-kernel time table from a rocprofv3 sqlite results db
(rocpd_* tables are UUID-suffixed and kernel names are interned in
rocpd_string): `python tools/kstats.py <results.db> [top_n]`."""
import sqlite3
import sys


def kstats(db_path, top=26):
    db = sqlite3.connect(db_path)
    names = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = next(n for n in names if n.startswith("rocpd_kernel_dispatch")
               )[len("rocpd_kernel_dispatch_"):]
    # display_name is plain TEXT in this schema (ROCm 7.2) and
    # dispatch.kernel_id == symbol.id
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6 AS ms
    FROM rocpd_kernel_dispatch_{sfx} k
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY ms DESC LIMIT ?"""
    rows = db.execute(q, (top,)).fetchall()
    total = db.execute(f"SELECT SUM(end-start)/1e6 "
                       f"FROM rocpd_kernel_dispatch_{sfx}").fetchone()[0]
    out = []
    for n, cnt, ms in rows:
        n = n.split("(")[0][:78]
        out.append(f"{ms:9.1f} ms {100*ms/total:5.1f}% {cnt:6d}  {n}")
    out.append(f"all-kernel total {total:.1f} ms")
    return "\n".join(out)


if __name__ == "__main__":
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 26
    print(kstats(sys.argv[1], top))
