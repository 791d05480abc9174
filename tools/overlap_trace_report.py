#!/usr/bin/env python3
"""Summarize comp/comm overlap from a rocprofv3 CSV kernel trace of ONE
overlap_bench worker.

Method: the comm stream carries the exchange's producer/staging work
(gather kernels + blit copies); a >2 ms gap between consecutive
comm-stream ops is a transport window (gloo CPU all-to-all in flight,
or RCCL DMA). Overlap evidence = default-stream kernels (the central
aggregation) executing INSIDE those windows. The comm stream is
auto-detected as the non-default stream carrying gather + copyBuffer
ops.

Usage: python tools/overlap_trace_report.py <trace_dir> [skip_s]
(skip_s: drop the first N seconds — setup/warmup; default 0)
"""
import csv
import glob
import os
import sys
from collections import Counter


def main():
    d = sys.argv[1]
    skip_s = float(sys.argv[2]) if len(sys.argv) > 2 else 0.0
    ks = []
    for fn in glob.glob(os.path.join(d, '*kernel_trace.csv')):
        for r in csv.DictReader(open(fn)):
            ks.append((int(r['Start_Timestamp']), int(r['End_Timestamp']),
                       r['Stream_Id'], r['Kernel_Name']))
    if not ks:
        raise SystemExit(f'no kernel trace CSVs under {d}')
    ks.sort()
    t0 = ks[0][0]
    lo = t0 + int(skip_s * 1e9)
    # comm stream: the non-zero stream with the most ops
    streams = Counter(q for _, _, q, _ in ks if q != '0')
    if not streams:
        raise SystemExit('no side-stream ops found (overlap path not taken?)')
    comm_id = streams.most_common(1)[0][0]
    comm = [(s, e) for s, e, q, _ in ks if q == comm_id and s > lo]
    dflt = [(s, e, n) for s, e, q, n in ks if q == '0' and s > lo]
    windows = []
    for (s1, e1), (s2, e2) in zip(comm, comm[1:]):
        if s2 - e1 > 2e6:
            windows.append((e1, s2))
    tot = sum(b - a for a, b in windows)
    busy = 0
    bykern = {}
    for s, e, n in dflt:
        for a, b in windows:
            o = min(e, b) - max(s, a)
            if o > 0:
                busy += o
                key = n.split('<')[0][:50]
                bykern[key] = bykern.get(key, 0) + o
    print(f'comm stream = {comm_id}; {len(windows)} transport windows, '
          f'total {tot/1e6:.1f} ms')
    print(f'default-stream kernel busy inside windows: {busy/1e6:.2f} ms '
          f'({100*busy/max(tot,1):.0f}% of window time covered by compute)')
    for k, v in sorted(bykern.items(), key=lambda x: -x[1])[:8]:
        print(f'  {v/1e6:8.2f} ms  {k}')


if __name__ == '__main__':
    main()
