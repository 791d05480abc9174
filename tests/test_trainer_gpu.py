"""End-to-end Trainer on the GPU: the full main.py path (config load,
partition cache, engine, assigner, model, train loop, checkpoints,
exp/ artifacts) on cuda:0 — integration coverage beyond the
engine-level GPU tests."""
import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_main_cli_gpu(tmp_path):
    env = dict(os.environ, MASTER_PORT='29561')
    subprocess.run(
        [sys.executable, 'graph_partition.py', '--dataset', 'reddit',
         '--partition_size', '1', '--scale', '0.01',
         '--partition_dir', str(tmp_path / 'parts')],
        cwd=REPO, check=True, capture_output=True, timeout=300)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'main.py'), '--dataset', 'reddit',
         '--model_name', 'sage', '--mode', 'AdaQP', '--assign_scheme',
         'uniform', '--num_epochs', '5', '--log_steps', '1', '--dtype',
         'bf16', '--partition_dir', str(tmp_path / 'parts'),
         '--exp_dir', str(tmp_path / 'exp'),
         '--ckpt_path', str(tmp_path / 'ck.pt'), '--ckpt_every', '2'],
        cwd=str(tmp_path), env=env, capture_output=True, text=True,
        timeout=600)
    assert out.returncode == 0, (out.stdout[-500:], out.stderr[-2000:])
    assert 'best:' in out.stdout
    assert (tmp_path / 'ck.pt').exists()
    mdir = tmp_path / 'exp' / 'reddit' / '1part' / 'sage' / 'metrics'
    assert (mdir / 'AdaQP_uniform_metrics.txt').exists()
