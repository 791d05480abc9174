"""Metric recorder (reference parity: ``AdaQP/util/recorder.py``)."""
from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch


class Recorder:
    def __init__(self):
        self.rows: List[List[float]] = []     # [train, val, test] per eval
        self.epochs: List[int] = []           # training epoch of each eval

    def add(self, metrics: Dict[str, float], epoch: Optional[int] = None):
        """Record one evaluation. ``epoch`` is the TRAINING epoch it was
        computed at (evals may be sparse with eval_every > 1); defaults to
        the row index for dense evaluation."""
        self.epochs.append(len(self.rows) if epoch is None else int(epoch))
        self.rows.append([metrics['train'], metrics['val'], metrics['test']])

    def best(self) -> Dict[str, float]:
        if not self.rows:
            return {'epoch': -1, 'val': 0.0, 'test': 0.0}
        t = torch.tensor(self.rows)
        i = int(t[:, 1].argmax())
        return {'epoch': self.epochs[i], 'val': float(t[i, 1]),
                'test': float(t[i, 2])}

    def save(self, out_dir: str, tag: str, extra: Optional[Dict] = None):
        os.makedirs(out_dir, exist_ok=True)
        b = self.best()
        with open(os.path.join(out_dir, f'{tag}_metrics.txt'), 'w') as f:
            f.write(f'best_epoch {b["epoch"]}\nbest_val {b["val"]:.4f}\n'
                    f'best_test {b["test"]:.4f}\n')
            for k, v in (extra or {}).items():
                f.write(f'{k} {v}\n')
        torch.save(torch.tensor(self.rows) if self.rows else torch.empty(0, 3),
                   os.path.join(out_dir, f'{tag}_val_curve.pt'))
