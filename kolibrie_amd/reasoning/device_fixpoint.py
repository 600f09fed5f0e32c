"""Driver for the K6 persistent single-workgroup device fixpoint
(ops/csrc/kernels.hip small_fixpoint).

Small working sets with chain-shaped derivations (deep taxonomy: 10 000
rounds of 1-fact deltas) are launch-latency-bound on the columnar path and
were routed to a HOST hash fixpoint in round 1.  The persistent kernel
runs the whole fixpoint in ONE launch — rounds are __syncthreads()
boundaries inside a single workgroup — so the "GPU number is a GPU
number" (no host round trips at all).

Eligible programs (else return None and the caller falls back):
  - rules with 1-2 positive premises, no filters, no negation;
  - every premise/conclusion triple is (Variable, Constant, Variable)
    with distinct s/o variables;
  - 2-premise rules share exactly ONE variable between the premises;
  - <=2 conclusions per rule, conclusion vars bound in the body.
This covers the BASELINE reasoning shapes (transitive closure, type
propagation / deep taxonomy, ancestor programs).

Ref semantics: datalog semi_naive.rs:17-86.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..storage.terms import Constant, TriplePattern, Variable
from .rule import Rule

MAX_ROUNDS = 200_000


def _vco(pat: TriplePattern):
    """(s_var_name, pred_i32, o_var_name) or None if not eligible."""
    if not (isinstance(pat.s, Variable) and isinstance(pat.p, Constant)
            and isinstance(pat.o, Variable)):
        return None
    if pat.s.name == pat.o.name:
        return None
    return pat.s.name, pat.p.id, pat.o.name


def _encode_rules(rules: List[Rule]):
    """Returns (rules_rows, pred_ids, adj_need, dispatch) or None."""
    pred_idx = {}

    def pid(i32: int) -> int:
        if i32 not in pred_idx:
            pred_idx[i32] = len(pred_idx)
        return pred_idx[i32]

    rows = []
    for rule in rules:
        if rule.filters or rule.negative_premise:
            return None
        if not (1 <= len(rule.premise) <= 2):
            return None
        if not (1 <= len(rule.conclusion) <= 2):
            return None
        p1 = _vco(rule.premise[0])
        if p1 is None:
            return None
        src = {p1[0]: 0, p1[2]: 1}
        if len(rule.premise) == 2:
            p2 = _vco(rule.premise[1])
            if p2 is None:
                return None
            shared = ({p1[0], p1[2]} & {p2[0], p2[2]})
            if len(shared) != 1:
                return None
            sv = next(iter(shared))
            j1 = 0 if p1[0] == sv else 1
            j2 = 0 if p2[0] == sv else 1
            for name, code in ((p2[0], 2), (p2[2], 3)):
                src.setdefault(name, code)
            kind, q1, q2 = 1, pid(p1[1]), pid(p2[1])
        else:
            kind, q1, q2, j1, j2 = 0, pid(p1[1]), 0, 0, 0
        row = [kind, q1, j1, q2, j2, len(rule.conclusion)]
        for c in rule.conclusion:
            cv = _vco(c)
            if cv is None or cv[0] not in src or cv[2] not in src:
                return None
            row += [pid(cv[1]), src[cv[0]], src[cv[2]]]
        while len(row) < 12:
            row += [0, 0, 0]
        rows.append(row[:12])
    n_preds = len(pred_idx)
    adj_need = [[0, 0] for _ in range(n_preds)]
    disp = [[] for _ in range(n_preds)]
    for ri, row in enumerate(rows):
        kind, q1, j1, q2, j2 = row[:5]
        disp[q1].append((ri, 0))
        if kind == 1:
            disp[q2].append((ri, 1))
            adj_need[q2][j2] = 1   # delta=premise1 probes premise2 by j2
            adj_need[q1][j1] = 1   # delta=premise2 probes premise1 by j1
    pred_ids = [0] * n_preds
    for i32, idx in pred_idx.items():
        pred_ids[idx] = i32
    return rows, pred_ids, adj_need, disp


def try_device_fixpoint(rules: List[Rule], facts, db) -> Optional[int]:
    """Run the whole fixpoint on-device; returns the number of derived
    facts, or None when the program/device is out of scope."""
    device = facts.device
    if device.type != "cuda" or not rules:
        return None
    enc = _encode_rules(rules)
    if enc is None:
        return None
    from ..ops import native_for
    native = native_for(facts.s)
    if native is None:
        return None
    rows, pred_ids, adj_need, disp = enc
    n_preds = len(pred_ids)
    n_disp = sum(len(d) for d in disp)
    if len(rows) > 64 or n_preds > 16 or n_disp > 128:
        return None  # exceeds the kernel's LDS metadata caps
    # map fact predicates to indexes; drop facts whose predicate no rule
    # mentions (they cannot derive or be derived)
    p = facts.p
    pidx = torch.full_like(p, -1)
    for idx, i32 in enumerate(pred_ids):
        pidx = torch.where(p == i32, torch.full_like(p, idx), pidx)
    keep = pidx >= 0
    fs = facts.s[keep].contiguous()
    fp = pidx[keep].contiguous()
    fo = facts.o[keep].contiguous()
    n_init = fs.numel()
    # start with tables small enough to stay L2-resident (the per-round
    # probe chain is the whole cost on 10k-round chain workloads); retry
    # bigger on overflow
    budget = max(65_536, 4 * n_init)

    rules_t = torch.tensor(rows, dtype=torch.int32).view(-1, 12)
    need_t = torch.tensor(adj_need, dtype=torch.int32).view(n_preds, 2)
    off, drule, dside = [0], [], []
    for q in range(n_preds):
        for ri, side in disp[q]:
            drule.append(ri)
            dside.append(side)
        off.append(len(drule))
    off_t = torch.tensor(off, dtype=torch.int32)
    drule_t = torch.tensor(drule or [0], dtype=torch.int32)
    dside_t = torch.tensor(dside or [0], dtype=torch.int32)

    for _attempt in range(3):
        out_s, out_p, out_o, seeded, overflow, rounds = native.small_fixpoint(
            rules_t, need_t, off_t, drule_t, dside_t, fs, fp, fo,
            budget, MAX_ROUNDS)
        if not overflow:
            break
        budget *= 16
    if overflow or rounds >= MAX_ROUNDS:
        return None  # fall back to the columnar / host paths
    facts.k6_rounds = rounds  # engagement marker (tests/profiling)
    n_out = out_s.numel()
    derived = n_out - seeded
    if derived > 0:
        pred_map = torch.tensor(pred_ids, dtype=torch.int32, device=device)
        ds = out_s[seeded:]
        dp = pred_map[out_p[seeded:].to(torch.long)]
        do_ = out_o[seeded:]
        facts.add_columns(ds.contiguous(), dp.contiguous(), do_.contiguous())
    return int(derived)
