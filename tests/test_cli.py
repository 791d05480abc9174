"""CLI surface smoke tests (single process, tiny synthetic graphs)."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_graph_partition_cli(tmp_path):
    out = subprocess.run(
        [sys.executable, 'graph_partition.py', '--dataset', 'reddit',
         '--partition_size', '2', '--scale', '0.002',
         '--partition_dir', str(tmp_path)],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-1500:]
    assert 'saved 2 partitions' in out.stdout
    d = os.path.join(str(tmp_path), 'reddit', '2part')
    assert os.path.exists(os.path.join(d, 'part0.pt'))
    assert os.path.exists(os.path.join(d, 'part1.pt'))
    meta = json.load(open(os.path.join(d, 'reddit.json')))
    assert meta['num_parts'] == 2


def test_main_cli_runs(tmp_path):
    env = dict(os.environ, MASTER_PORT='29532')
    # tiny scale via the partition cache: pre-partition at scale 0.002
    subprocess.run(
        [sys.executable, 'graph_partition.py', '--dataset', 'reddit',
         '--partition_size', '1', '--scale', '0.002',
         '--partition_dir', str(tmp_path / 'parts')],
        cwd=REPO, check=True, capture_output=True, timeout=300)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'main.py'), '--dataset', 'reddit',
         '--model_name', 'sage', '--mode', 'AdaQP-q',
         '--assign_scheme', 'uniform', '--num_epochs', '3', '--log_steps', '1',
         '--partition_dir', str(tmp_path / 'parts')],
        cwd=str(tmp_path), env=env, capture_output=True, text=True,
        timeout=600)
    assert out.returncode == 0, (out.stdout[-800:], out.stderr[-1500:])
    assert 'best:' in out.stdout
    exp = tmp_path / 'exp' / 'reddit' / '1part' / 'sage'
    assert (exp / 'metrics').is_dir()
    assert (exp / 'time').is_dir()


def test_checkpoint_resume(tmp_path):
    env = dict(os.environ, MASTER_PORT='29533')
    subprocess.run(
        [sys.executable, 'graph_partition.py', '--dataset', 'reddit',
         '--partition_size', '1', '--scale', '0.002',
         '--partition_dir', str(tmp_path / 'parts')],
        cwd=REPO, check=True, capture_output=True, timeout=300)
    ck = str(tmp_path / 'ckpt.pt')
    base = [sys.executable, os.path.join(REPO, 'main.py'), '--dataset',
            'reddit', '--model_name', 'gcn', '--mode', 'Vanilla',
            '--log_steps', '1', '--partition_dir', str(tmp_path / 'parts'),
            '--ckpt_path', ck, '--ckpt_every', '2']
    out = subprocess.run(base + ['--num_epochs', '4'], cwd=str(tmp_path),
                         env=env, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-1500:]
    assert os.path.exists(ck)
    out = subprocess.run(base + ['--num_epochs', '6', '--resume'],
                         cwd=str(tmp_path), env=env, capture_output=True,
                         text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-1500:]
    assert 'resumed from' in out.stdout
