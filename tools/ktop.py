#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db into a per-kernel time table.

rocprofv3 --kernel-trace --stats writes a sqlite database under the -d
directory (``<pid>_results.db``); this prints (and optionally writes) the
per-kernel call count / total / average / share table the committed
profiles/* summaries use, with the mangled display names reduced to the
``k_*`` kernel identifiers.

Usage: python tools/ktop.py gpurun_out/profd/runc/163_results.db [out.txt]
"""

import re
import sqlite3
import sys


def summarize(db_path: str):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sym = next(t for t in tables if "info_kernel_symbol" in t)
    dis = next(t for t in tables if "kernel_dispatch" in t)
    rows = cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end - d.start)/1e6,
               AVG(d.end - d.start)/1e6
        FROM {dis} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC
    """).fetchall()
    total = sum(r[2] for r in rows) or 1.0

    def short(name):
        m = re.search(r"k_[a-z0-9_]+", name)
        return (m.group(0) if m else name)[:46]

    lines = [f"{'kernel':46s} {'calls':>6s} {'total_ms':>10s} "
             f"{'avg_ms':>9s} {'%kern':>6s}"]
    for name, cnt, tot, avg in rows:
        lines.append(f"{short(name):46s} {cnt:6d} {tot:10.3f} "
                     f"{avg:9.4f} {100 * tot / total:6.1f}")
    lines.append("")
    lines.append(f"{'TOTAL kernel time':46s} {'':6s} {total:10.3f}")
    return "\n".join(lines) + "\n"


if __name__ == "__main__":
    if len(sys.argv) < 2:
        sys.exit(__doc__)
    text = summarize(sys.argv[1])
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text)
    print(text)
