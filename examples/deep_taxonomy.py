#!/usr/bin/env python3
"""Deep-taxonomy reasoning (ref: examples/sparql_syntax/knowledge_graph/
deep_taxonomy.rs — type propagation over an N-level subclass chain,
20-iteration timing loop)."""
import sys, time
sys.path.insert(0, ".")
from scripts.bench_reasoning import build_deep_taxonomy

DEPTH = 1000
times = []
for i in range(20):
    r = build_deep_taxonomy(DEPTH, "cpu")
    t0 = time.perf_counter()
    n = r.infer_new_facts_semi_naive()
    times.append(time.perf_counter() - t0)
print(f"depth {DEPTH}: avg {sum(times)/len(times)*1000:.1f} ms "
      f"min {min(times)*1000:.1f} ms ({n} inferences/iter)")
