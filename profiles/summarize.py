#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db (kernel-trace) into a markdown table.

Usage: python profiles/summarize.py gpurun_out/prof/runc/NNN_results.db > profiles/out.md
"""

import sqlite3
import sys


def summarize(path, top=25):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
               AVG(k.end-k.start)/1e3
        FROM {kd} k JOIN {ks} ks ON k.kernel_id=ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {top}""").fetchall()
    a, b = cur.execute(f"SELECT MIN(start),MAX(end) FROM {kd}").fetchone()
    total = cur.execute(
        f"SELECT SUM(end-start)/1e6 FROM {kd}").fetchone()[0]
    print(f"| total ms | calls | avg us | kernel |")
    print(f"|---:|---:|---:|---|")
    for name, n, tot, avg in rows:
        short = name.split("(")[0][:100]
        print(f"| {tot:.1f} | {n} | {avg:.1f} | `{short}` |")
    print()
    print(f"- kernel time total: {total:.0f} ms; "
          f"kernel-span wall: {(b - a) / 1e9:.2f} s")


if __name__ == "__main__":
    summarize(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 25)
