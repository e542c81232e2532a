#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db into a small text report (top kernels
by total time; with --pmc, per-kernel counter aggregates)."""

import glob
import sqlite3
import sys


def main():
    db_glob = sys.argv[1]
    out_path = sys.argv[2]
    files = sorted(glob.glob(db_glob))
    with open(out_path, 'w') as out:
        for f in files:
            db = sqlite3.connect(f)
            out.write('== %s ==\n' % f)
            try:
                for r in db.execute(
                        "SELECT name, total_calls, total_duration, "
                        "average, percentage FROM top_kernels LIMIT 40"):
                    out.write('%7.2f%% %7d calls avg %9.1fus tot '
                              '%10.0fus  %s\n'
                              % (r[4], r[1], r[3], r[2], r[0][:110]))
            except Exception as e:
                out.write('no top_kernels: %r\n' % (e,))
            # PMC counters if present
            try:
                tables = [t[0] for t in db.execute(
                    "SELECT name FROM sqlite_master WHERE type IN "
                    "('table','view')")]
                ct = [t for t in tables if 'counter' in t.lower()]
                out.write('counter tables: %s\n' % ct[:6])
                for t in ct:
                    cols = [c[1] for c in db.execute(
                        'PRAGMA table_info(%s)' % t)]
                    out.write('  %s cols: %s\n' % (t, cols[:12]))
                # common rocpd layout: rocpd_counter joins dispatch
                if 'counters_collection' in tables:
                    q = ("SELECT kernel_name, counter_name, SUM(value) "
                         "FROM counters_collection GROUP BY kernel_name, "
                         "counter_name ORDER BY 3 DESC LIMIT 60")
                    for r in db.execute(q):
                        out.write('%14.0f  %-24s %s\n'
                                  % (r[2], r[1], r[0][:80]))
            except Exception as e:
                out.write('counter dump failed: %r\n' % (e,))
            db.close()


if __name__ == '__main__':
    main()
