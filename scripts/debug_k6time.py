import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scripts.bench_reasoning import build_deep_taxonomy
from kolibrie_amd.reasoning.device_fixpoint import try_device_fixpoint

for depth in (1000, 10000):
    r = build_deep_taxonomy(depth, "cuda:0")
    r._flush()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    d = try_device_fixpoint(r.rules, r.facts, r.db)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) * 1000
    print(f"depth {depth}: kernel-path total {dt:.1f} ms, derived {d}, "
          f"rounds {getattr(r.facts, 'k6_rounds', None)}")
