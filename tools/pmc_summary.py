"""Summarize a rocprofv3 --pmc counter CSV: per-kernel-family mean of
MfmaUtil / VALUBusy / OccupancyPercent."""
import csv
import sys
from collections import defaultdict

FAMS = ['attn_fwd', 'attn_bwd_dq', 'attn_bwd_dkv', 'pcgemm',
        'linear_gemm', 'geglu_fwd', 'geglu_bwd', 'layernorm_fwd',
        'layernorm_bwd', 'gatemul', 'attn_delta', 'pairrep', 'Cijk']


def fam(name):
    for f in FAMS:
        if f in name:
            return f
    return None


def main(path):
    acc = defaultdict(lambda: defaultdict(list))
    with open(path) as fh:
        for row in csv.DictReader(fh):
            f = fam(row['Kernel_Name'])
            if f:
                acc[f][row['Counter_Name']].append(
                    float(row['Counter_Value']))
    print(f"{'kernel':14s} {'MfmaUtil':>9s} {'VALUBusy':>9s} {'Occup%':>8s}"
          f" {'n':>5s}")
    for f in FAMS:
        if f not in acc:
            continue
        c = acc[f]
        def m(k):
            v = c.get(k, [])
            return sum(v) / len(v) if v else float('nan')
        n = len(c.get('MfmaUtil', []))
        print(f'{f:14s} {m("MfmaUtil"):9.1f} {m("VALUBusy"):9.1f} '
              f'{m("OccupancyPercent"):8.1f} {n:5d}')


if __name__ == '__main__':
    main(sys.argv[1])
