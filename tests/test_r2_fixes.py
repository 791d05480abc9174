"""Regression tests for the round-1 review findings (VERDICT.md #7,
ADVICE.md): recorder epoch indexing with sparse evaluation, partition
cache scale validation, trainer log location, and the fused-SAGE gate
inspecting the right tensor."""
import json
import os
import subprocess
import sys

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_recorder_best_epoch_with_sparse_eval():
    """With eval_every > 1 the recorder must report the TRAINING epoch of
    the best eval, not the dense row index (VERDICT weak #6)."""
    from adaqp_amd.runtime.recorder import Recorder
    r = Recorder()
    # evals at epochs 0, 5, 10 (eval_every=5); best val at epoch 5
    r.add({'train': .1, 'val': .2, 'test': .3}, epoch=0)
    r.add({'train': .4, 'val': .9, 'test': .8}, epoch=5)
    r.add({'train': .5, 'val': .6, 'test': .7}, epoch=10)
    b = r.best()
    assert b['epoch'] == 5
    assert abs(b['test'] - .8) < 1e-6  # fp32 storage


def test_recorder_default_epoch_is_row_index():
    from adaqp_amd.runtime.recorder import Recorder
    r = Recorder()
    for i in range(3):
        r.add({'train': .0, 'val': float(i), 'test': .5})
    assert r.best()['epoch'] == 2


def test_scale_mismatch_cache_is_rejected(tmp_path):
    """An explicit --scale that disagrees with the cached partition's
    scale must fail loudly instead of silently reusing it."""
    env = dict(os.environ, MASTER_PORT='29537')
    subprocess.run(
        [sys.executable, 'graph_partition.py', '--dataset', 'reddit',
         '--partition_size', '1', '--scale', '0.002',
         '--partition_dir', str(tmp_path / 'parts')],
        cwd=REPO, check=True, capture_output=True, timeout=300)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'main.py'), '--dataset', 'reddit',
         '--model_name', 'gcn', '--mode', 'Vanilla', '--num_epochs', '1',
         '--scale', '0.004', '--partition_dir', str(tmp_path / 'parts')],
        cwd=str(tmp_path), env=env, capture_output=True, text=True,
        timeout=300)
    assert out.returncode != 0
    assert 'built at scale=0.002' in (out.stderr + out.stdout)


def test_trainer_log_written_under_exp_not_cwd(tmp_path):
    env = dict(os.environ, MASTER_PORT='29538')
    subprocess.run(
        [sys.executable, 'graph_partition.py', '--dataset', 'reddit',
         '--partition_size', '1', '--scale', '0.002',
         '--partition_dir', str(tmp_path / 'parts')],
        cwd=REPO, check=True, capture_output=True, timeout=300)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'main.py'), '--dataset', 'reddit',
         '--model_name', 'gcn', '--mode', 'Vanilla', '--num_epochs', '2',
         '--partition_dir', str(tmp_path / 'parts')],
        cwd=str(tmp_path), env=env, capture_output=True, text=True,
        timeout=600)
    assert out.returncode == 0, out.stderr[-1500:]
    assert not (tmp_path / 'trainer.log').exists()
    assert (tmp_path / 'exp' / 'logs' / 'trainer.log').exists()


def test_fused_gate_checks_x_self(monkeypatch):
    """DistSAGEConv must gate the fused path on x_self (the tensor the
    kernel actually consumes), not h_neigh twice (VERDICT weak #5)."""
    import adaqp_amd.models.sage as sage_mod
    seen = {}

    def spy_ok(x, h, n_out):
        seen['x_cols'] = x.shape[1]
        seen['h_cols'] = h.shape[1]
        return False

    monkeypatch.setattr(sage_mod, 'fused_dual_linear_ok', spy_ok)

    class FakeGraph:
        num_inner = 4

    class FakeEngine:
        compute_dtype = torch.bfloat16
        graph = FakeGraph()

    conv = sage_mod.DistSAGEConv(6, 16, layer=0, aggregator_type='mean')
    h_neigh = torch.randn(4, 6)
    monkeypatch.setattr(sage_mod, 'dist_aggregate',
                        lambda x, e, layer, training: h_neigh)
    x = torch.randn(7, 6)  # inner + remote rows; x_self = x[:4]
    conv.forward(FakeEngine(), x)
    assert seen['x_cols'] == 6 and seen['h_cols'] == 6


def test_bench_metric_label_world_accurate(tmp_path):
    """bench.py must label the metric by the ACTUAL world size
    (VERDICT weak #1): a 1-rank run must not claim '8-part'."""
    env = dict(os.environ, MASTER_PORT='29539')
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'bench.py'), '--gpus', '1',
         '--steps', '2', '--warmup', '1', '--cpu', '--scale', '0.002',
         '--part-dir', str(tmp_path / 'parts')],
        cwd=str(tmp_path), env=env, capture_output=True, text=True,
        timeout=600)
    assert out.returncode == 0, out.stderr[-1500:]
    line = [l for l in out.stdout.splitlines() if l.startswith('{')][-1]
    rec = json.loads(line)
    assert '1-part' in rec['metric']
    assert '8-part' not in rec['metric']
    assert rec['n_gpus'] == 1
