"""Driver-contract regression: `python bench.py --gpus N --steps K
--warmup W` prints exactly one JSON line with the required keys."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ['metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup',
            'ms_per_step', 'higher_is_better', 'scaling', 'vs_baseline',
            'dtype', 'data', 'config']


def _run(args):
    env = dict(os.environ)
    env.pop('WORLD_SIZE', None)
    env.pop('RANK', None)
    out = subprocess.run(
        [sys.executable, 'bench.py'] + args, cwd=ROOT, env=env,
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.strip().startswith('{')]
    assert len(lines) == 1, out.stdout
    return json.loads(lines[0])


def test_single_process_json_contract():
    d = _run(['--gpus', '1', '--steps', '2', '--warmup', '1',
              '--model', 'lenet', '--dataset', 'cifar10',
              '--batch-size', '4', '--dtype', 'fp32'])
    for k in REQUIRED:
        assert k in d, k
    assert d['n_gpus'] == 1
    assert d['steps'] == 2 and d['warmup'] == 1
    assert d['value'] > 0 and d['ms_per_step'] > 0
    assert d['higher_is_better'] is True
    assert d['scaling'] == 'weak'
    assert d['data'] == 'synthetic'
    assert d['config']['global_batch'] == 4
    assert d['config']['parallelism'] == 'dp1'


def test_merge_arms_accepted():
    for arm in ('wfbp', 'single', 'threshold:1000'):
        d = _run(['--gpus', '1', '--steps', '1', '--warmup', '0',
                  '--model', 'lenet', '--dataset', 'cifar10',
                  '--batch-size', '4', '--dtype', 'fp32',
                  '--merge', arm])
        assert d['config']['merge'] == arm
