"""Find the op that breaks hipGraph capture (full traceback)."""
import torch
from alphafold2_amd import Alphafold2
from alphafold2_amd.data import synthetic_batch
from alphafold2_amd.utils import get_bucketed_distance_matrix

m = Alphafold2(dim=64, depth=2, heads=2, dim_head=64).cuda().train()
opt = torch.optim.Adam(m.parameters(), lr=3e-4, capturable=True, foreach=True)
b = synthetic_batch(1, 64, 8, device="cuda", seed=0)
tgt = get_bucketed_distance_matrix(b["coords"], b["mask"])

def step():
    opt.zero_grad(set_to_none=False)
    with torch.autocast("cuda", dtype=torch.bfloat16, cache_enabled=False):
        ret = m(b["seq"], b["msa"], mask=b["mask"], msa_mask=b["msa_mask"])
        loss = torch.nn.functional.cross_entropy(
            ret.distance.permute(0, 3, 1, 2).float(), tgt,
            ignore_index=-100) + ret.msa_mlm_loss.float()
    loss.backward()
    opt.step()
    return loss

s = torch.cuda.Stream()
s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        step()
torch.cuda.current_stream().wait_stream(s)
torch.cuda.synchronize()

g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    loss = step()
print("captured OK")
g.replay()
torch.cuda.synchronize()
print("replay OK, loss:", loss.item())
