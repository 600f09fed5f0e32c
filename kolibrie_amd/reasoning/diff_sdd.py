"""Differentiable WMC: per-variable gradients of the weighted model count.

Ref parity: shared/src/diff_sdd.rs:15 (wmc_gradient — the circuit gradient
that backpropagates query probability into neural seed weights for
neurosymbolic training, execute_ml_train.rs).

Assumes probability-normalized weights (w+ = p, w- = 1-p), so the
marginalization spans of unconstrained variables are constant 1 and the
gradient flows only through constrained decisions:
    dWMC/dp_v = sum over v-decision nodes of adjoint * (WMC(hi) - WMC(lo)).
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict

from .sdd import FALSE, TRUE, SddManager


def wmc_gradient(manager: SddManager, node: int) -> Dict[int, float]:
    """Returns {var: dWMC/dp_var} for the circuit rooted at `node`."""
    # forward pass: node values
    value: Dict[int, float] = {TRUE: 1.0, FALSE: 0.0}

    def fwd(nid: int) -> float:
        if nid in value:
            return value[nid]
        var, hi, lo = manager.nodes[nid]
        v = manager.pos_weight.get(var, 1.0) * fwd(hi) \
            + manager.neg_weight.get(var, 0.0) * fwd(lo)
        value[nid] = v
        return v

    fwd(node)

    # reverse pass: adjoints in topological (descending id ~ creation) order
    adjoint: Dict[int, float] = defaultdict(float)
    adjoint[node] = 1.0
    grads: Dict[int, float] = defaultdict(float)
    reachable = set()

    def mark(nid: int):
        if nid in reachable or nid <= TRUE:
            return
        reachable.add(nid)
        _, hi, lo = manager.nodes[nid]
        mark(hi)
        mark(lo)

    mark(node)
    for nid in sorted(reachable, reverse=True):
        a = adjoint[nid]
        if a == 0.0:
            continue
        var, hi, lo = manager.nodes[nid]
        wp = manager.pos_weight.get(var, 1.0)
        wn = manager.neg_weight.get(var, 0.0)
        # d node / d p_var = value(hi) - value(lo)   (w- = 1 - w+)
        grads[var] += a * (value.get(hi, 0.0) - value.get(lo, 0.0))
        if hi > TRUE:
            adjoint[hi] += a * wp
        if lo > TRUE:
            adjoint[lo] += a * wn
    return dict(grads)
