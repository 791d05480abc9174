#!/usr/bin/env python3
"""Render exp/ artifacts into the Vanilla-vs-AdaQP comparison table the
reference README centers on (``/root/reference/README.md:136-138``).

Walks ``<root>/<dataset>/<P>part/<model>/metrics/<tag>_metrics.txt``
(written by Trainer.save) and prints one markdown table per
(dataset, partition count, model): best test accuracy, mean epoch time,
and speedup vs that group's Vanilla run.

Usage: python tools/results_table.py [--root exp] [--dataset reddit]
"""
import argparse
import os
import sys


def parse_metrics(path):
    out = {}
    with open(path) as f:
        for line in f:
            parts = line.split()
            if len(parts) == 2:
                out[parts[0]] = parts[1]
    return out


def collect(root, dataset=None):
    rows = []
    if not os.path.isdir(root):
        return rows
    for ds in sorted(os.listdir(root)):
        if dataset and ds != dataset:
            continue
        dpath = os.path.join(root, ds)
        if not os.path.isdir(dpath):
            continue
        for part in sorted(os.listdir(dpath)):
            if not part.endswith('part'):
                continue
            for model in sorted(os.listdir(os.path.join(dpath, part))):
                mdir = os.path.join(dpath, part, model, 'metrics')
                if not os.path.isdir(mdir):
                    continue
                for fn in sorted(os.listdir(mdir)):
                    if not fn.endswith('_metrics.txt'):
                        continue
                    tag = fn[:-len('_metrics.txt')]
                    m = parse_metrics(os.path.join(mdir, fn))
                    rows.append({'dataset': ds, 'parts': part, 'model': model,
                                 'tag': tag, **m})
    return rows


def render(rows, out=sys.stdout):
    groups = {}
    for r in rows:
        groups.setdefault((r['dataset'], r['parts'], r['model']), []).append(r)
    for (ds, part, model), rs in sorted(groups.items()):
        vanilla = next((r for r in rs if r['tag'] == 'Vanilla'), None)
        vt = (float(vanilla['mean_epoch_time_s'])
              if vanilla and vanilla.get('mean_epoch_time_s', 'n/a') != 'n/a'
              else None)
        print(f'\n## {ds} {part} {model}', file=out)
        print('| mode | best test acc | epoch time (s) | speedup vs Vanilla |',
              file=out)
        print('|---|---|---|---|', file=out)
        for r in rs:
            t = r.get('mean_epoch_time_s', 'n/a')
            sp = 'n/a'
            if vt and t != 'n/a' and float(t) > 0:
                sp = f'{vt / float(t):.2f}x'
            print(f"| {r['tag']} | {float(r.get('best_test', 0)):.4f} "
                  f"| {t} | {sp} |", file=out)


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--root', default='exp')
    p.add_argument('--dataset', default=None)
    args = p.parse_args()
    rows = collect(args.root, args.dataset)
    if not rows:
        raise SystemExit(f'no metrics found under {args.root}')
    render(rows)


if __name__ == '__main__':
    main()
