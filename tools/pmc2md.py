"""Summarise a rocprofv3 --pmc counter_collection.csv by kernel.

Usage: python tools/pmc2md.py <csv-or-db> [name-filter]
Handles both the csv output and the rocpd sqlite db (pmc_events view).
"""

import sys


def from_db(path, filt):
    import sqlite3
    db = sqlite3.connect(path)
    cur = db.cursor()
    q = """
    select k.name, p.counter_name, sum(p.value), count(distinct p.dispatch_id)
    from pmc_events p join kernels k on k.dispatch_id = p.dispatch_id
    group by k.name, p.counter_name"""
    try:
        rows = cur.execute(q).fetchall()
    except Exception:
        # schema variant: counter info joined differently
        rows = cur.execute("""
          select s.string, i.name, sum(p.value), count(*)
          from rocpd_pmc_event p
          join rocpd_info_pmc i on p.pmc_id = i.id
          join rocpd_kernel_dispatch k on p.event_id = k.id
          join rocpd_info_kernel_symbol ks on k.kernel_id = ks.id
          join rocpd_string s on ks.display_name = s.id
          group by s.string, i.name""").fetchall()
    agg = {}
    for name, cname, val, nd in rows:
        if filt and filt not in name:
            continue
        agg.setdefault(name[:80], {})[cname] = (val, nd)
    for name, counters in agg.items():
        print(f"\n## {name}")
        for cname, (val, nd) in sorted(counters.items()):
            print(f"  {cname}: total {val:.3e} over {nd} dispatches")


def from_csv(path, filt):
    import csv
    agg = {}
    with open(path) as f:
        r = csv.DictReader(f)
        for row in r:
            name = (row.get("Kernel_Name") or row.get("kernel_name") or "")
            if filt and filt not in name:
                continue
            c = row.get("Counter_Name") or row.get("counter_name")
            v = float(row.get("Counter_Value") or row.get("counter_value")
                      or 0)
            d = agg.setdefault(name[:80], {})
            tot, n = d.get(c, (0.0, 0))
            d[c] = (tot + v, n + 1)
    for name, counters in agg.items():
        print(f"\n## {name}")
        for cname, (val, n) in sorted(counters.items()):
            print(f"  {cname}: total {val:.3e} over {n} samples")


if __name__ == "__main__":
    path = sys.argv[1]
    filt = sys.argv[2] if len(sys.argv) > 2 else ""
    if path.endswith(".db"):
        from_db(path, filt)
    else:
        from_csv(path, filt)
