#!/usr/bin/env python3
"""Extract a per-kernel time summary from a rocprofv3 rocpd sqlite DB
(rocprofv3 --kernel-trace writes <out>_results.db; tables are suffixed
per-process).  Prints one JSON object: top kernels by total time."""

import json
import sqlite3
import sys


def main():
    db = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 12
    con = sqlite3.connect(db)
    tabs = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    out = []
    for t in tabs:
        sfx = t[len("rocpd_kernel_dispatch_"):]
        rows = con.execute(
            f"SELECT ks.kernel_name, COUNT(*), SUM(k.end-k.start)/1e6, "
            f"AVG(k.end-k.start)/1e3 "
            f"FROM rocpd_kernel_dispatch_{sfx} k "
            f"JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id "
            f"GROUP BY ks.kernel_name ORDER BY 3 DESC LIMIT {top}"
        ).fetchall()
        out.append({"process": sfx, "kernels": [
            {"name": r[0][:70], "calls": r[1],
             "total_ms": round(r[2], 3), "avg_us": round(r[3], 2)}
            for r in rows]})
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
