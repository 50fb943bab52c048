#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db (kernel stats) into a markdown table.

Usage: python tools/summarize_prof.py gpurun_out/profN/runc/*_results.db [out.md]
"""
import sqlite3
import sys


def main():
    db_path = sys.argv[1]
    out = sys.argv[2] if len(sys.argv) > 2 else None
    cur = sqlite3.connect(db_path).cursor()
    rows = cur.execute(
        "SELECT name, total_calls, total_duration, average, percentage "
        "FROM top_kernels LIMIT 40").fetchall()
    lines = ["| % | calls | total_us | avg_us | kernel |", "|---|---|---|---|---|"]
    for name, calls, tot, avg, pct in rows:
        short = (name if len(name) < 110 else name[:107] + "...").replace("|", "\\|")
        lines.append(f"| {pct:.2f} | {calls} | {tot:.0f} | {avg:.1f} | `{short}` |")
    text = "\n".join(lines) + "\n"
    if out:
        open(out, "w").write(text)
    else:
        print(text)


if __name__ == "__main__":
    main()
