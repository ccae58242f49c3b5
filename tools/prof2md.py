"""Summarise a rocprofv3 results.db (rocpd SQLite) into a markdown
kernel table for profiles/.

Usage: python tools/prof2md.py <results.db> [title...]
"""

import sqlite3
import sys


def main():
    db_path = sys.argv[1]
    title = " ".join(sys.argv[2:]) or db_path
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    rows = cur.execute(
        "select name, count(*) c, sum(duration)/1e6 ms from kernels "
        "group by name order by ms desc limit 30").fetchall()
    tot, n = cur.execute(
        "select sum(duration)/1e6, count(*) from kernels").fetchone()
    print(f"# rocprofv3 kernel stats — {title}")
    print()
    print(f"Total GPU kernel time: {tot/1e3:.3f} s across {n} dispatches.")
    print()
    print("| % | total ms | calls | kernel |")
    print("|---|---|---|---|")
    for name, c, ms in rows:
        short = name.replace("|", "\\|")[:110]
        print(f"| {100*ms/tot:.1f}% | {ms:.1f} | {c} | `{short}` |")


if __name__ == "__main__":
    main()
