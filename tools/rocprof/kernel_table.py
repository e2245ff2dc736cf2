#!/usr/bin/env python3
"""Summarize a rocprofv3 --kernel-trace results db into a per-kernel table.

Usage: python tools/rocprof/kernel_table.py <results.db> [steps]
Writes "<kernel> calls total_ms avg_us" sorted by total time; if steps is
given, also prints ms/step. (ROCm 7.2's rocprofv3 writes the stats into a
rocpd sqlite db; the csv exporter does not always fire under gpurun.)
"""
import sqlite3
import sys


def main():
    db = sys.argv[1]
    steps = float(sys.argv[2]) if len(sys.argv) > 2 else None
    c = sqlite3.connect(db)
    tabs = [r[0] for r in c.execute(
        "select name from sqlite_master where name like 'rocpd_kernel_dispatch%'")]
    if not tabs:
        sys.exit("no kernel dispatch table in " + db)
    suf = tabs[0].replace("rocpd_kernel_dispatch_", "")
    rows = c.execute(f"""
        select s.display_name, count(*), sum(k.end-k.start)/1e6,
               avg(k.end-k.start)/1e3
        from rocpd_kernel_dispatch_{suf} k
        join rocpd_info_kernel_symbol_{suf} s on k.kernel_id = s.id
        group by 1 order by 3 desc""").fetchall()
    tot = sum(r[2] for r in rows)
    hdr = f"{'kernel':70s} {'calls':>6s} {'tot_ms':>9s} {'avg_us':>8s} {'pct':>5s}"
    if steps:
        hdr += f" {'ms/step':>8s}"
    print(hdr)
    for n, cnt, ms, us in rows:
        line = f"{n[:70]:70s} {cnt:6d} {ms:9.2f} {us:8.1f} {100*ms/tot:5.1f}"
        if steps:
            line += f" {ms/steps:8.2f}"
        print(line)
    print(f"TOTAL {tot:.1f} ms" + (f" ({tot/steps:.1f} ms/step)" if steps else ""))


if __name__ == "__main__":
    main()
