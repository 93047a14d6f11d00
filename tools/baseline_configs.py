"""Measure BASELINE.json configs 3-5 on one MI355X (driver runs DP=8).

Runs bench.py as subprocesses with the per-config flags and prints one
JSON line per config.  Step counts are small — the structure modules
run eager fp32 (K7 open) and config 5 is crop 512 / msa 512.
"""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

CONFIGS = {
    'cfg3_reversible_dim384': [
        '--dim', '384', '--depth', '12', '--reversible',
        '--batch', '4', '--steps', '5', '--warmup', '2'],
    'cfg4_se3_crop384': [
        '--structure-module', 'se3', '--crop-len', '384',
        '--msa-depth', '128', '--batch', '1', '--steps', '3',
        '--warmup', '1'],
    'cfg5_egnn_crop512_msa512': [
        '--structure-module', 'egnn', '--predict-angles',
        '--crop-len', '512', '--msa-depth', '512', '--batch', '1',
        '--steps', '3', '--warmup', '1'],
}


def main():
    for name, flags in CONFIGS.items():
        r = subprocess.run(
            [sys.executable, 'bench.py'] + flags,
            cwd=ROOT, capture_output=True, text=True, timeout=1200)
        line = ''
        for l in reversed(r.stdout.strip().splitlines()):
            if l.startswith('{'):
                line = l
                break
        print(f'### {name} rc={r.returncode}')
        if line:
            print(line, flush=True)
        else:
            print(r.stderr[-1500:], flush=True)


if __name__ == '__main__':
    main()
