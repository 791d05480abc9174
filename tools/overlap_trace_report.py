#!/usr/bin/env python3
"""Summarize comp/comm overlap from a rocprofv3 CSV trace of ONE
overlap_bench worker (kernel-trace + memory-copy-trace).

The gloo-staged exchange is: D2H copy (comm stream) -> CPU all-to-all
(GPU idle on the comm stream) -> H2D copy. True overlap means compute
kernels execute INSIDE those [D2H end, H2D start] windows — that is
exactly the time the decomposed path hides. Reports total window time
and the kernel busy time inside it.

Usage: python tools/overlap_trace_report.py <trace_dir>
"""
import csv
import glob
import os
import sys


def load_rows(pattern):
    rows = []
    for fn in glob.glob(pattern):
        with open(fn) as f:
            rows.extend(csv.DictReader(f))
    return rows


def col(row, *names):
    for n in names:
        for k in row:
            if k.strip('"').lower() == n.lower():
                return row[k].strip('"')
    raise KeyError(f'{names} not in {list(row)[:12]}')


def main():
    d = sys.argv[1]
    kern = load_rows(os.path.join(d, '*kernel_trace.csv'))
    copies = load_rows(os.path.join(d, '*memory_copy_trace.csv'))
    if not kern or not copies:
        raise SystemExit(f'no kernel/copy trace CSVs under {d}')
    ks = [(int(col(r, 'Start_Timestamp')), int(col(r, 'End_Timestamp')),
           col(r, 'Kernel_Name', 'Name')) for r in kern]
    cs = sorted((int(col(r, 'Start_Timestamp')), int(col(r, 'End_Timestamp')),
                 col(r, 'Direction', 'Name', 'Kind')) for r in copies)
    # exchange windows: D2H end -> next H2D start
    windows = []
    for i, (s, e, dirn) in enumerate(cs):
        dl = dirn.lower()
        if 'device_to_host' in dl or 'd2h' in dl or 'devicetohost' in dl:
            for s2, e2, d2 in cs[i + 1:]:
                d2l = d2.lower()
                if 'host_to_device' in d2l or 'h2d' in d2l or 'hosttodevice' in d2l:
                    if s2 > e:
                        windows.append((e, s2))
                    break
    # merge overlapping windows
    windows.sort()
    merged = []
    for w in windows:
        if merged and w[0] <= merged[-1][1]:
            merged[-1] = (merged[-1][0], max(merged[-1][1], w[1]))
        else:
            merged.append(w)
    total_win = sum(b - a for a, b in merged)
    busy = 0
    by_kernel = {}
    for s, e, name in ks:
        for a, b in merged:
            o = min(e, b) - max(s, a)
            if o > 0:
                busy += o
                key = name.split('<')[0].split('(')[0]
                by_kernel[key] = by_kernel.get(key, 0) + o
    print(f'{len(merged)} exchange windows, total {total_win/1e6:.2f} ms')
    print(f'kernel busy time inside windows: {busy/1e6:.2f} ms '
          f'({100.0*busy/max(total_win,1):.0f}% of window time overlapped '
          f'with compute)')
    for k, v in sorted(by_kernel.items(), key=lambda x: -x[1])[:8]:
        print(f'  {v/1e6:8.2f} ms  {k}')


if __name__ == '__main__':
    main()
