"""Summarize a rocprofv3 rocpd SQLite database into a kernel-stats
markdown table.

rocprofv3 on ROCm 7 writes <name>_results.db; this extracts (a) the
whole-run top kernels and (b) a steady-state window ending at the last
dispatch (to exclude MIOpen find-mode probe kernels, which dominate the
whole-run totals but never run in the replayed graph).

Usage: python benchmarks/rocpd_summary.py <results.db> <window_s> <out.md>
"""
import re
import sqlite3
import sys


def shorten(n):
    n = n.replace('void ', '').replace('(anonymous namespace)::', '')
    m = re.match(r'([A-Za-z_0-9:]+)', n)
    return (m.group(1) if m else n)[:72]


def table(cur, t0, limit=25):
    rows = list(cur.execute(
        'SELECT name, COUNT(*), SUM(duration) FROM kernels '
        'WHERE start >= ? GROUP BY name', (t0,)))
    agg = {}
    for name, cnt, dur in rows:
        s = shorten(name)
        c, d = agg.get(s, (0, 0))
        agg[s] = (c + cnt, d + dur)
    tot = sum(d for _, d in agg.values())
    out = ['| kernel | calls | total ms | us/call | % |',
           '|---|---|---|---|---|']
    for s, (c, d) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:limit]:
        out.append('| %s | %d | %.2f | %.1f | %.2f |'
                   % (s.replace('|', '/'), c, d / 1e6, d / c / 1e3,
                      100 * d / tot))
    return out, tot


def main():
    db_path, window_s, out_path = (sys.argv[1], float(sys.argv[2]),
                                   sys.argv[3])
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tmax, tmin = cur.execute(
        'SELECT MAX(end), MIN(start) FROM kernels').fetchone()
    lines = ['# Kernel stats (%s)' % db_path.split('/')[-1], '']
    t_all, tot_all = table(cur, 0)
    t0 = tmax - int(window_s * 1e9)
    t_win, tot_win = table(cur, t0)
    lines.append('## Steady state — last %.2f s (timed window; graph '
                 'replay, excludes warmup + MIOpen find probes)'
                 % window_s)
    lines.append('GPU busy in window: %.1f ms (%.1f%%)'
                 % (tot_win / 1e6, 100 * tot_win / (window_s * 1e9)))
    lines.append('')
    lines.extend(t_win)
    lines.append('')
    lines.append('## Whole run (includes MIOpen/CK find-mode probes '
                 'during warmup — NOT steady state)')
    lines.append('total GPU kernel time: %.1f s over %.1f s wall'
                 % (tot_all / 1e9, (tmax - tmin) / 1e9))
    lines.append('')
    lines.extend(t_all)
    report = '\n'.join(lines)
    with open(out_path, 'w') as f:
        f.write(report + '\n')
    print(report[:2000])


if __name__ == '__main__':
    main()
