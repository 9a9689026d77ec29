#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results .db into per-kernel counter totals.

Per MI355X_MICROARCH.md §HBM: FETCH_SIZE on gfx950 reports HALF the bytes of
a wide coalesced streaming read (double before comparing with a byte count);
WRITE_SIZE is uncalibrated — calibrate on a known pattern.  Counter values
land in KB units in the db (rocprofv3 convention for *_SIZE).

Usage: python tools/pmcsum.py <results.db> [out.txt]
"""

import re
import sqlite3
import sys


def summarize(db_path: str):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    lines = []
    try:
        sym = next(t for t in tables if "info_kernel_symbol" in t)
        dis = next(t for t in tables if "kernel_dispatch" in t)
        pmc = next(t for t in tables if "counters_collection" in t or
                   "pmc_event" in t or "counter" in t.lower())
    except StopIteration:
        return "tables: " + ", ".join(tables) + "\n"
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({pmc})")]
    lines.append(f"# pmc table {pmc}: {cols}")
    # common rocprofv3 schema: counters_collection(dispatch_id, counter_id/
    # name, value); join through kernel_dispatch -> symbol
    try:
        name_col = "counter_name" if "counter_name" in cols else "name"
        rows = cur.execute(f"""
            SELECT s.display_name, p.{name_col}, COUNT(DISTINCT d.id),
                   SUM(p.value), AVG(p.value)
            FROM {pmc} p
            JOIN {dis} d ON p.dispatch_id = d.dispatch_id
            JOIN {sym} s ON d.kernel_id = s.id
            GROUP BY s.display_name, p.{name_col}
            ORDER BY SUM(p.value) DESC
        """).fetchall()
    except sqlite3.OperationalError as e:
        return f"schema mismatch ({e}); tables: {tables}\ncols: {cols}\n"

    def short(name):
        m = re.search(r"k_[a-zA-Z0-9_]+", name)
        return (m.group(0) if m else name)[:44]

    for name, counter, n, tot, avg in rows:
        lines.append(f"{short(name):44s} {counter:12s} n={n:4d} "
                     f"total={tot:.6g} per-dispatch={avg:.6g}")
    return "\n".join(lines) + "\n"


if __name__ == "__main__":
    if len(sys.argv) < 2:
        sys.exit(__doc__)
    text = summarize(sys.argv[1])
    print(text)
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text)
