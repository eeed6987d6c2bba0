"""Summarize rocprofv3 output (kernel-trace .db and/or --pmc
counter_collection CSVs) into small per-kernel tables — run ON the GPU
box so only the summaries travel back through gpurun_out.

Usage: python tests/summarize_prof.py <dir> [<dir> ...]
Prints one table per directory to stdout.
"""
import csv
import glob
import os
import sqlite3
import sys
from collections import defaultdict


def summarize_db(path):
    db = sqlite3.connect(path)
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = [t for t in tabs if t.startswith('rocpd_kernel_dispatch_')]
    sym = [t for t in tabs if t.startswith('rocpd_info_kernel_symbol_')]
    if not disp or not sym:
        return
    q = f"""SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
            AVG(d.end-d.start)/1e6
            FROM {disp[0]} d JOIN {sym[0]} s ON d.kernel_id = s.id
            GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC
            LIMIT 24"""
    print(f"{'total_ms':>10} {'calls':>6} {'ms/call':>10}  kernel")
    for name, n, tot, avg in db.execute(q):
        print(f"{tot:10.2f} {n:6d} {avg:10.4f}  {name[:78]}")


def summarize_pmc(files):
    acc = defaultdict(lambda: defaultdict(float))
    calls = defaultdict(lambda: defaultdict(int))
    for f in files:
        with open(f) as fh:
            r = csv.DictReader(fh)
            for row in r:
                k = row.get('Kernel_Name') or row.get('Kernel-Name', '')
                c = row.get('Counter_Name') or row.get('Counter-Name', '')
                v = float(row.get('Counter_Value')
                          or row.get('Counter-Value') or 0)
                k = k.replace('void ', '').replace(
                    '(anonymous namespace)::', '')
                acc[k.split('(')[0][:70]][c] += v
                calls[k.split('(')[0][:70]][c] += 1
    for k in sorted(acc, key=lambda k: -max(acc[k].values())):
        for c, v in acc[k].items():
            n = calls[k][c]
            # FETCH/WRITE_SIZE report in KB per the guide
            print(f"{c:>12} disp={n:4d} total_GB={v*1024/1e9:10.3f} "
                  f"GB/disp={v*1024/1e9/max(1,n):8.3f}  {k}")


def main():
    for d in sys.argv[1:]:
        print(f"==== {d} ====")
        dbs = glob.glob(os.path.join(d, '**', '*.db'), recursive=True)
        for p in dbs:
            summarize_db(p)
        csvs = glob.glob(os.path.join(d, '**', '*counter_collection*.csv'),
                         recursive=True)
        if csvs:
            summarize_pmc(csvs)
        if not dbs and not csvs:
            print('(nothing found; contents: %s)'
                  % os.listdir(d)[:10])


if __name__ == '__main__':
    main()
