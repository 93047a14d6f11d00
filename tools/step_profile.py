"""Op-level attribution of one training step (torch.profiler with
shapes): identifies which ATen ops emit the add/reduce/cat kernels and
which model sites own the big Tensile GEMMs.  Run on a GPU box."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.profiler import ProfilerActivity, profile

from alphafold2_amd import Alphafold2
from alphafold2_amd.data import synthetic_batch
from alphafold2_amd.utils import get_bucketed_distance_matrix


def main():
    torch.manual_seed(0)
    model = Alphafold2(dim=256, depth=12, heads=8, dim_head=64).cuda()
    model.train()
    opt = torch.optim.Adam(model.parameters(), lr=3e-4, fused=True)
    b = synthetic_batch(2, 256, 128, device='cuda', seed=0)
    tgt = get_bucketed_distance_matrix(b['coords'], b['mask'])

    def step():
        opt.zero_grad(set_to_none=True)
        with torch.autocast('cuda', dtype=torch.bfloat16):
            ret = model(b['seq'], b['msa'], mask=b['mask'],
                        msa_mask=b['msa_mask'])
            loss = torch.nn.functional.cross_entropy(
                ret.distance.permute(0, 3, 1, 2).float(), tgt,
                ignore_index=-100)
            loss = loss + ret.msa_mlm_loss.float()
        loss.backward()
        opt.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof:
        for _ in range(2):
            step()
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by='self_cuda_time_total', row_limit=55, max_src_column_width=60))


if __name__ == '__main__':
    main()
