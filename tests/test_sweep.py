"""Sweep scripts + comparison table: one command runs
the modes x models sweep and renders the Vanilla-vs-AdaQP table the
reference README shows."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_reddit_all_sweep_tiny(tmp_path):
    """scripts/reddit_all.sh end-to-end at tiny scale: 2 parts, gcn,
    Vanilla + AdaQP-q, then the table over the produced exp/ tree."""
    env = dict(os.environ,
               PARTS='2', MODELS='gcn', MODES='Vanilla AdaQP-q',
               PORT='29551', EXP=f'{tmp_path}/exp',
               EXTRA=f'--scale 0.002 --num_epochs 3 --log_steps 1 '
                     f'--partition_dir {tmp_path}/parts '
                     f'--exp_dir {tmp_path}/exp')
    out = subprocess.run(['bash', 'scripts/reddit_all.sh'], cwd=REPO, env=env,
                         capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, (out.stdout[-800:], out.stderr[-2000:])
    mdir = tmp_path / 'exp' / 'reddit' / '2part' / 'gcn' / 'metrics'
    assert (mdir / 'Vanilla_metrics.txt').exists()
    assert (mdir / 'AdaQP-q_adaptive_metrics.txt').exists()
    table = subprocess.run(
        [sys.executable, 'tools/results_table.py', '--root',
         str(tmp_path / 'exp')],
        cwd=REPO, capture_output=True, text=True, timeout=120)
    assert table.returncode == 0, table.stderr[-1000:]
    assert '## reddit 2part gcn' in table.stdout
    assert 'Vanilla' in table.stdout and 'AdaQP-q_adaptive' in table.stdout
    assert 'speedup vs Vanilla' in table.stdout


def test_results_table_rendering(tmp_path):
    """Table math: speedup column is vanilla_time / mode_time."""
    mdir = tmp_path / 'yelp' / '4part' / 'sage' / 'metrics'
    os.makedirs(mdir)
    (mdir / 'Vanilla_metrics.txt').write_text(
        'best_epoch 10\nbest_val 0.9000\nbest_test 0.8000\n'
        'mean_epoch_time_s 0.4000\ntotal_time_s 40.0\n')
    (mdir / 'AdaQP_adaptive_metrics.txt').write_text(
        'best_epoch 12\nbest_val 0.8990\nbest_test 0.7980\n'
        'mean_epoch_time_s 0.2000\ntotal_time_s 20.0\n')
    out = subprocess.run(
        [sys.executable, 'tools/results_table.py', '--root', str(tmp_path)],
        cwd=REPO, capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert '2.00x' in out.stdout
    assert '0.7980' in out.stdout
