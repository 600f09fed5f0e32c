#!/usr/bin/env python3
"""Anytime probabilistic inference with escalation (ref:
kolibrie/benches/hybrid_probability.rs, shared/src/hybrid.rs)."""
import sys
sys.path.insert(0, ".")
from kolibrie_amd.reasoning.hybrid import (
    HybridConfig, evaluate_hybrid, materialize_lineage,
)
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable

P, Q = 100, 101
rule = Rule(
    premise=[TriplePattern(Variable("x"), Constant(P), Variable("y")),
             TriplePattern(Variable("y"), Constant(P), Variable("z"))],
    conclusion=[TriplePattern(Variable("x"), Constant(Q), Variable("z"))],
)
seeds = {(1, P, 2): 0.9, (2, P, 3): 0.8, (1, P, 4): 0.5, (4, P, 3): 0.7}
store, nodes, weights = materialize_lineage([rule], seeds)
res = evaluate_hybrid(store, nodes[(1, Q, 3)], weights,
                      HybridConfig(threshold=0.5))
print(f"P(1 -Q-> 3) = {res.probability:.4f} status={res.status} "
      f"above_threshold={res.above_threshold} k={res.metrics.k_used}")
