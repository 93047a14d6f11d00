"""Merge TunableOp tuning CSVs into the shipped gfx950 cache.

Usage:  python tools/merge_tunableop.py extra1.csv [extra2.csv ...]

Keeps the master's Validator block and unions the Gemm entries (first
occurrence wins, so the master's measured solutions take precedence).
"""
import os
import sys

MASTER = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), 'alphafold2_amd', 'runtime',
    'tunableop_gfx950.csv')


def load(path):
    validators, entries = [], {}
    with open(path) as f:
        for line in f:
            line = line.rstrip('\n')
            if not line:
                continue
            if line.startswith('Validator'):
                validators.append(line)
            else:
                key = line.split(',')[1] if line.count(',') >= 2 else line
                entries.setdefault(key, line)
    return validators, entries


def main(extras):
    validators, master = load(MASTER)
    before = len(master)
    for path in extras:
        _, entries = load(path)
        for k, v in entries.items():
            master.setdefault(k, v)
    with open(MASTER, 'w') as f:
        f.write('\n'.join(validators) + '\n')
        f.write('\n'.join(master.values()) + '\n')
    print(f'{MASTER}: {before} -> {len(master)} gemm entries')


if __name__ == '__main__':
    main(sys.argv[1:])
