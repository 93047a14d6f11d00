"""Driver-contract tests for bench.py: single-process CPU run and the
torchrun multi-rank launch path (gloo backend, 127.0.0.1 rendezvous)."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TINY = ['--dim', '32', '--depth', '1', '--crop-len', '32',
        '--msa-depth', '4', '--batch', '1', '--dtype', 'fp32',
        '--steps', '2', '--warmup', '1']


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(('127.0.0.1', 0))
        return s.getsockname()[1]


def _parse_last_json(out):
    for line in reversed(out.strip().splitlines()):
        if line.startswith('{'):
            return json.loads(line)
    raise AssertionError(f'no JSON line in output:\n{out}')


@pytest.mark.timeout(300)
def test_bench_single_process_cpu():
    out = subprocess.run(
        [sys.executable, 'bench.py'] + TINY,
        cwd=ROOT, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    d = _parse_last_json(out.stdout)
    assert d['n_gpus'] == 1
    assert d['value'] > 0
    assert d['config']['global_batch'] == 1
    assert d['metric'].startswith('training samples/sec')
    assert d['scaling'] == 'weak'
    assert d['data'] == 'synthetic'


@pytest.mark.timeout(300)
def test_bench_torchrun_two_ranks_cpu():
    env = dict(os.environ)
    env['MASTER_ADDR'] = '127.0.0.1'
    out = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', str(_free_port()), 'bench.py', '--gpus', '2'] + TINY,
        cwd=ROOT, capture_output=True, text=True, timeout=280, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    d = _parse_last_json(out.stdout)
    assert d['n_gpus'] == 2
    assert d['config']['global_batch'] == 2
    assert d['config']['parallelism'] == 'dp2'
