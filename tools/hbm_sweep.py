"""288 GB HBM3E sizing sweep (BASELINE config 3: reversible dim=384).

Walks batch / crop / msa upward under the reversible trunk and records
peak HBM per config plus step time, stopping gracefully at the first
OOM (caught, never crashes the box).  Run on a GPU box:

    python tools/hbm_sweep.py [--dim 384] [--depth 12]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def try_config(dim, depth, batch, crop, msa_depth, reversible, steps=2):
    from alphafold2_amd import Alphafold2
    from alphafold2_amd.data import synthetic_batch
    from alphafold2_amd.utils import get_bucketed_distance_matrix

    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    try:
        model = Alphafold2(dim=dim, depth=depth, heads=8, dim_head=64,
                           max_seq_len=max(2048, crop),
                           reversible=reversible).cuda().train()
        opt = torch.optim.Adam(model.parameters(), lr=3e-4, foreach=True)
        b = synthetic_batch(batch, crop, msa_depth, device='cuda', seed=0)
        tgt = get_bucketed_distance_matrix(b['coords'], b['mask'])
        times = []
        for i in range(steps + 1):
            t0 = time.perf_counter()
            opt.zero_grad(set_to_none=True)
            with torch.autocast('cuda', dtype=torch.bfloat16):
                ret = model(b['seq'], b['msa'], mask=b['mask'],
                            msa_mask=b['msa_mask'])
                loss = torch.nn.functional.cross_entropy(
                    ret.distance.permute(0, 3, 1, 2).float(), tgt,
                    ignore_index=-100)
                if ret.msa_mlm_loss is not None:
                    loss = loss + ret.msa_mlm_loss.float()
            loss.backward()
            opt.step()
            torch.cuda.synchronize()
            if i > 0:
                times.append(time.perf_counter() - t0)
        peak = torch.cuda.max_memory_allocated() / 2**30
        ms = sum(times) / len(times) * 1000
        sps = batch / (ms / 1000)
        return peak, ms, sps
    except torch.cuda.OutOfMemoryError:
        return None
    finally:
        for n in ('model', 'opt', 'b', 'tgt', 'ret', 'loss'):
            if n in locals():
                del locals()[n]
        torch.cuda.empty_cache()


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--dim', type=int, default=384)
    p.add_argument('--depth', type=int, default=12)
    args = p.parse_args()

    total = torch.cuda.get_device_properties(0).total_memory / 2**30
    print(f'# device HBM: {total:.0f} GiB', flush=True)

    # axis 1: batch at flagship crop/msa, reversible vs standard
    for reversible in (False, True):
        tag = 'rev' if reversible else 'std'
        for batch in (5, 8, 12, 16, 24, 32):
            r = try_config(args.dim, args.depth, batch, 256, 128, reversible)
            if r is None:
                print(f'{tag} dim={args.dim} b={batch} crop=256 msa=128: OOM',
                      flush=True)
                break
            peak, ms, sps = r
            print(f'{tag} dim={args.dim} b={batch} crop=256 msa=128: '
                  f'peak={peak:.1f} GiB  {ms:.0f} ms/step  '
                  f'{sps:.2f} samples/s', flush=True)

    # axis 2: crop/msa growth at batch 1 (long-context lever)
    for crop, msa in ((384, 192), (512, 256), (768, 384), (1024, 512)):
        r = try_config(args.dim, args.depth, 1, crop, msa, True)
        if r is None:
            print(f'rev dim={args.dim} b=1 crop={crop} msa={msa}: OOM',
                  flush=True)
            break
        peak, ms, sps = r
        print(f'rev dim={args.dim} b=1 crop={crop} msa={msa}: '
              f'peak={peak:.1f} GiB  {ms:.0f} ms/step  '
              f'{sps:.2f} samples/s', flush=True)


if __name__ == '__main__':
    main()
