#!/usr/bin/env python
"""Dump top-kernel stats from a rocprofv3 rocpd sqlite db (gpurun_out/prof/...)
into a text summary for profiles/."""
import sqlite3
import sys


def main(db_path, out_path=None, n=30):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    rows = cur.execute(
        "SELECT name, total_calls, total_duration, average, percentage "
        f"FROM top_kernels LIMIT {int(n)}").fetchall()
    lines = [f"{'%':>6}  {'total ms':>10}  {'calls':>6}  {'avg us':>9}  name"]
    for name, calls, total_us, avg_us, pct in rows:
        lines.append(f"{pct:6.2f}  {total_us / 1e3:10.2f}  {calls:6d}  "
                     f"{avg_us:9.1f}  {name[:110]}")
    text = "\n".join(lines) + "\n"
    if out_path:
        open(out_path, "w").write(text)
    else:
        print(text)


if __name__ == "__main__":
    main(*sys.argv[1:])
