"""Summarize a rocprofv3 rocpd .db (kernel stats + PMC counters) into a
small text table — run ON the GPU box so only the summary travels back.

Usage: python scripts/pmc_summarize.py results.db [top_n]
"""
import re
import sqlite3
import sys
from collections import defaultdict


def main():
    db = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 15
    c = sqlite3.connect(db)
    tabs = [r[0] for r in c.execute(
        "select name from sqlite_master where type='table'")]
    u = [t for t in tabs if t.startswith('rocpd_kernel_dispatch_')][0] \
        .replace('rocpd_kernel_dispatch_', '')

    total = c.execute(f"select sum(end-start)/1e6 from "
                      f"rocpd_kernel_dispatch_{u}").fetchone()[0]
    print(f"total kernel ms: {total:.1f}")
    rows = list(c.execute(f"""
        select s.display_name, count(*), sum(d.end-d.start)/1e6,
               avg(d.end-d.start)/1e3, d.kernel_id
        from rocpd_kernel_dispatch_{u} d
        join rocpd_info_kernel_symbol_{u} s on d.kernel_id=s.id
        group by s.display_name order by 3 desc limit {top}"""))
    for name, n, ms, avg_us, kid in rows:
        nm = re.sub(r'\(.*', '', name)[:60]
        print(f"{ms:9.2f} ms {100*ms/total:5.1f}% n={n:6d} "
              f"avg={avg_us:8.2f}us  {nm}")

    pmc_t = [t for t in tabs if t.startswith('rocpd_pmc_event_')]
    if not pmc_t:
        return
    cols = [r[1] for r in c.execute(f"pragma table_info({pmc_t[0]})")]
    print("\nPMC per kernel (sum over dispatches):")
    # rocpd schema: pmc_event(pmc_id -> info_pmc, event_id -> dispatch event)
    try:
        q = f"""
        select s.display_name, p.name, sum(e.value)
        from {pmc_t[0]} e
        join rocpd_info_pmc_{u} p on e.pmc_id = p.id
        join rocpd_kernel_dispatch_{u} d on e.event_id = d.event_id
        join rocpd_info_kernel_symbol_{u} s on d.kernel_id = s.id
        group by s.display_name, p.name"""
        agg = defaultdict(dict)
        for name, cnt, val in c.execute(q):
            agg[re.sub(r'\(.*', '', name)[:48]][cnt] = val
        for name, d in sorted(agg.items()):
            print(f"  {name}: " + " ".join(
                f"{k}={v:.3e}" for k, v in sorted(d.items())))
    except sqlite3.OperationalError as e:
        print("pmc join failed:", e)
        print("pmc cols:", cols)


if __name__ == "__main__":
    main()
