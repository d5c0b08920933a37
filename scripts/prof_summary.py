#!/usr/bin/env python3
"""Summarize a rocprofv3 sqlite results db into a small text report
(top kernels by total time) — run ON the GPU box so only the summary ships
back through gpurun_out."""
import glob
import sqlite3
import sys


def summarize(db_path, out_path, top=30, window=0.0):
    """window <= 1: fraction of the timeline to SKIP from the start.
    window > 1: keep only the LAST `window` MILLISECONDS (robust steady-state
    selection when MIOpen-find dominates the wall)."""
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    if not tables:
        raise SystemExit('no kernel dispatch table in %s' % db_path)
    sfx = tables[0].replace('rocpd_kernel_dispatch_', '')
    t0, t1 = cur.execute(
        "SELECT MIN(start), MAX(end) FROM rocpd_kernel_dispatch_%s"
        % sfx).fetchone()
    if window > 1.0:
        cut = t1 - window * 1e6  # rocprof timestamps are ns
    else:
        cut = t0 + (t1 - t0) * window
    where = "WHERE CAST(k.start AS REAL) >= %d" % cut
    tot = cur.execute(
        "SELECT SUM(k.end-k.start)/1e6, COUNT(*), "
        "(MAX(k.end)-MIN(k.start))/1e6 "
        "FROM rocpd_kernel_dispatch_%s k %s" % (sfx, where)).fetchone()
    hdr = ('last %.0f ms' % window) if window > 1 else ('skip=%.0f%%' % (window * 100))
    lines = ['db: %s (window %s)' % (db_path, hdr),
             'TOTAL kernel %.1f ms / %d dispatches; wall span %.1f ms' % tot]
    q = ("SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6, "
         "AVG(k.end-k.start)/1e3 FROM rocpd_kernel_dispatch_%s k "
         "JOIN rocpd_info_kernel_symbol_%s ks ON k.kernel_id=ks.id %s "
         "GROUP BY ks.display_name ORDER BY SUM(k.end-k.start) DESC "
         "LIMIT %d" % (sfx, sfx, where, top))
    for name, calls, tot_ms, avg_us in cur.execute(q):
        lines.append('%9.2f ms %7dx %8.1fus  %s'
                     % (tot_ms, calls, avg_us, name[:100]))
    with open(out_path, 'w') as f:
        f.write('\n'.join(lines) + '\n')
    print('\n'.join(lines[:12]))


if __name__ == '__main__':
    pattern = sys.argv[1] if len(sys.argv) > 1 else 'gpurun_out/prof*/**/*.db'
    out = sys.argv[2] if len(sys.argv) > 2 else 'gpurun_out/prof_summary.txt'
    window = float(sys.argv[3]) if len(sys.argv) > 3 else 0.0
    dbs = sorted(glob.glob(pattern, recursive=True))
    if not dbs:
        raise SystemExit('no dbs matching %s' % pattern)
    summarize(dbs[-1], out, window=window)
