#!/usr/bin/env python3
"""Inference CLI: sequence (or a3m MSA) -> 3D structure (PDB).

    python predict.py --seq MKTAYIAKQRQISFVKSHFSRQLEERLGLIEVQ --out pred.pdb
    python predict.py --a3m query.a3m --checkpoint ckpt.pt --out pred.pdb

Runs the Alphafold2 model (optionally from a trained checkpoint) with
predict_coords, optional recycling iterations, writes a CA-trace PDB
and per-residue confidence, optionally applies the native geometric
relaxer.
"""
import argparse

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument('--seq', type=str, default=None,
                   help='one-letter amino-acid sequence')
    p.add_argument('--a3m', type=str, default=None,
                   help='a3m MSA file (first record = query)')
    p.add_argument('--checkpoint', type=str, default=None)
    p.add_argument('--out', type=str, default='prediction.pdb')
    p.add_argument('--dim', type=int, default=256)
    p.add_argument('--depth', type=int, default=6)
    p.add_argument('--heads', type=int, default=8)
    p.add_argument('--dim-head', type=int, default=64)
    p.add_argument('--structure-module', type=str, default='ipa',
                   choices=['ipa', 'se3', 'egnn'])
    p.add_argument('--recycles', type=int, default=3)
    p.add_argument('--msa-depth', type=int, default=32)
    p.add_argument('--relax', action='store_true',
                   help='apply the native geometric relaxer')
    p.add_argument('--device', type=str, default=None)
    return p.parse_args()


def main():
    args = parse_args()
    from alphafold2_amd import Alphafold2
    from alphafold2_amd.data.trrosetta import encode_seq
    from alphafold2_amd.geometry.pdb import read_msa
    from alphafold2_amd.runtime import load_checkpoint

    assert args.seq or args.a3m, 'provide --seq or --a3m'

    device = torch.device(args.device) if args.device else (
        torch.device('cuda') if torch.cuda.is_available()
        else torch.device('cpu'))

    if args.a3m:
        records = read_msa(args.a3m, args.msa_depth)
        seqs = [s for _, s in records]
        seq = encode_seq(seqs[0])[None].to(device)
        msa = torch.stack([encode_seq(s) for s in seqs])[None].to(device)
    else:
        seq = encode_seq(args.seq)[None].to(device)
        msa = seq[:, None, :]

    model = Alphafold2(
        dim=args.dim, depth=args.depth, heads=args.heads,
        dim_head=args.dim_head, predict_coords=True,
        structure_module_type=args.structure_module,
    ).to(device).eval()

    if args.checkpoint:
        load_checkpoint(args.checkpoint, model, restore_rng=False)

    mask = torch.ones_like(seq).bool()
    msa_mask = torch.ones_like(msa).bool()

    amp = torch.autocast('cuda', dtype=torch.bfloat16) \
        if device.type == 'cuda' else torch.no_grad()
    recyclables = None
    with torch.no_grad(), amp:
        for _ in range(max(1, args.recycles)):
            coords, ret = model(seq, msa, mask=mask, msa_mask=msa_mask,
                                recyclables=recyclables,
                                return_aux_logits=True,
                                return_recyclables=True)
            recyclables = ret.recyclables
        coords, confidence = model(seq, msa, mask=mask, msa_mask=msa_mask,
                                   recyclables=recyclables,
                                   return_confidence=True)

    coords = coords.float().cpu()
    confidence = confidence.float().cpu()

    if args.relax:
        import importlib.util
        import os
        spec = importlib.util.spec_from_file_location(
            'refinement', os.path.join(os.path.dirname(
                os.path.abspath(__file__)), 'scripts', 'refinement.py'))
        ref = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(ref)
        coords = ref.relax_structure(coords)

    write_ca_pdb(seq[0].cpu(), coords[0], confidence[0, :, 0], args.out)
    print(f'wrote {args.out} '
          f'(mean confidence {confidence.sigmoid().mean().item():.3f})')


def write_ca_pdb(seq_ids, coords, confidence, path):
    """Minimal CA-trace PDB writer with confidence in the B-factor."""
    from alphafold2_amd.vocab import VOCAB, ONE_TO_THREE_LETTER_MAP
    lines = []
    conf = torch.sigmoid(confidence) * 100
    for i, (aa_id, xyz) in enumerate(zip(seq_ids.tolist(), coords.tolist())):
        aa = VOCAB._int2char[int(aa_id)]
        res3 = ONE_TO_THREE_LETTER_MAP.get(aa, 'UNK')
        x, y, z = xyz
        lines.append(
            f"ATOM  {i + 1:5d}  CA  {res3:>3s} A{i + 1:4d}    "
            f"{x:8.3f}{y:8.3f}{z:8.3f}  1.00{conf[i].item():6.2f}")
    lines.append("END")
    with open(path, 'w') as f:
        f.write("\n".join(lines) + "\n")
    return path


if __name__ == '__main__':
    main()
