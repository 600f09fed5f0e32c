import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from kolibrie_amd.ops import native_for

dev = "cuda:0"
# program: closure over pred 0: (x,0,y),(y,0,z)->(x,0,z)
rules = torch.tensor([[1, 0, 1, 0, 0, 1, 0, 0, 3, 0, 0, 0]], dtype=torch.int32)
need = torch.tensor([[1, 1]], dtype=torch.int32)
off = torch.tensor([0, 2], dtype=torch.int32)
drule = torch.tensor([0, 0], dtype=torch.int32)
dside = torch.tensor([0, 1], dtype=torch.int32)
fs = torch.tensor([1, 2, 3], dtype=torch.int32, device=dev)
fp = torch.tensor([0, 0, 0], dtype=torch.int32, device=dev)
fo = torch.tensor([2, 3, 4], dtype=torch.int32, device=dev)
native = native_for(fs)
out_s, out_p, out_o, seeded, overflow, rounds = native.small_fixpoint(
    rules, need, off, drule, dside, fs, fp, fo, 1000, 100)
print("seeded", seeded, "overflow", overflow, "rounds", rounds,
      "n_out", out_s.numel())
print(sorted(zip(out_s.cpu().tolist(), out_p.cpu().tolist(),
                 out_o.cpu().tolist())))
# expect derived: (1,0,3),(2,0,4),(1,0,4) -> 6 total facts
