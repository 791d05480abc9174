"""The driver contract: bench.py must run under torch.distributed.run
with N>1 (one rank per device — CPU here) and print one valid JSON line
with the required fields."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench(nproc, extra):
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           f'--nproc-per-node={nproc}', '--master-addr', '127.0.0.1',
           '--master-port', '29520', 'bench.py', '--cpu',
           '--gpus', str(nproc), '--steps', '2', '--warmup', '1',
           '--scale', '0.002'] + extra
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                         timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith('{')]
    assert len(lines) == 1, out.stdout
    return json.loads(lines[0])


def test_bench_json_contract_2rank(tmp_path):
    d = json.loads(json.dumps(_run_bench(
        2, ['--part-dir', str(tmp_path / 'parts')])))
    assert d['n_gpus'] == 2
    assert d['steps'] == 2 and d['warmup'] == 1
    assert d['higher_is_better'] is False
    assert d['scaling'] == 'strong'
    assert d['unit'] == 's/epoch'
    assert d['value'] > 0 and abs(d['ms_per_step'] - d['value'] * 1000) < 1e-6
    assert d['dtype'] == 'fp32'
    assert 'synthetic' in d['data']
    assert 'ogbn-products' in d['metric']
    assert 0.0 <= d['config']['test_acc'] <= 1.0


def test_bench_json_contract_1rank_defaults(tmp_path):
    cmd = [sys.executable, 'bench.py', '--cpu', '--steps', '2', '--warmup', '1',
           '--scale', '0.002', '--part-dir', str(tmp_path / 'parts')]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                         timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads([l for l in out.stdout.splitlines() if l.startswith('{')][0])
    assert d['n_gpus'] == 1


def test_bench_adaptive_scheme_2rank(tmp_path):
    d = _run_bench(2, ['--part-dir', str(tmp_path / 'parts'),
                       '--assign-scheme', 'adaptive'])
    assert d['config']['mode'] == 'AdaQP'
