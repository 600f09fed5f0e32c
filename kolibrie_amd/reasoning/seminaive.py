"""Datalog materialisation strategies: naive and semi-naive fixpoints.

Ref parity: datalog/src/materialisation/{my_naive.rs, semi_naive.rs:10-92,
semi_naive_parallel.rs:11-177, infer_generic.rs:9-53}.

MI355X-native redesign: facts are int32 (s,p,o) columns
with a sorted GraphIndex rebuilt per round; each round joins the delta
against rule premises with the same columnar probe/join machinery the query
engine uses (K1/K2 kernels on device), instantiates conclusions
column-wise, dedups with a sort-unique pass and subtracts known facts with
a membership mask — no per-row host work inside the loop.

The optional `tags` column (float32 per fact) carries semiring provenance
values; tag-improved facts re-enter the delta (ref
provenance_semi_naive.rs:185-197 delta_improved).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from ..engine.bindings import Bindings
from ..engine.scan import scan_probe, scan_unit
from ..engine.tensor_utils import membership_mask, unique_rows
from ..storage.dataset import GraphIndex
from ..storage.terms import Constant, TriplePattern, UNBOUND, Variable
from .rule import Rule


class FactStore:
    """Growing columnar fact set with a sorted index."""

    def __init__(self, device):
        self.device = torch.device(device)
        e = torch.empty(0, dtype=torch.int32, device=self.device)
        self.s, self.p, self.o = e, e.clone(), e.clone()
        self._index: Optional[GraphIndex] = None

    @property
    def n(self) -> int:
        return self.s.numel()

    def index(self) -> GraphIndex:
        if self._index is None:
            self._index = GraphIndex.from_columns(
                self.s, self.p, self.o, device=self.device, dedup=False)
        return self._index

    def add_columns(self, s, p, o) -> int:
        """Add new facts (must already be deduped against self); returns the
        number added."""
        if s.numel() == 0:
            return 0
        self.s = torch.cat([self.s, s])
        self.p = torch.cat([self.p, p])
        self.o = torch.cat([self.o, o])
        self._index = None
        return s.numel()

    def set_columns(self, s, p, o):
        self.s, self.p, self.o = s, p, o
        self._index = None

    def sorted_unique_rows(self):
        return unique_rows([self.s, self.p, self.o])


def _match_premise_against_delta(
    prem: TriplePattern, ds, dp, do_, device
) -> Optional[Bindings]:
    """Filter delta triples by the premise's constants; bind its variables.
    Returns None when no delta row matches."""
    n = ds.numel()
    mask = torch.ones(n, dtype=torch.bool, device=device)
    cols = (ds, dp, do_)
    var_first: Dict[str, torch.Tensor] = {}
    for i, t in enumerate(prem.terms()):
        if isinstance(t, Constant):
            mask &= cols[i] == t.id
        elif isinstance(t, Variable):
            if t.name in var_first:
                mask &= var_first[t.name] == cols[i]
            else:
                var_first[t.name] = cols[i]
        else:
            return None  # quoted patterns in rules: not in scope for K6
    if not bool(mask.any()):
        return None
    out_cols = {v: c[mask] for v, c in var_first.items()}
    return Bindings(out_cols, int(mask.sum().item()), device)


def _join_premise_all_facts(
    b: Bindings, prem: TriplePattern, idx: GraphIndex, device
) -> Bindings:
    """Join current bindings with a premise matched against ALL facts
    (ref rules.rs:167 join_premise_with_hash_join -> K1/K2)."""
    consts: Dict[int, int] = {}
    var_pos: Dict[int, str] = {}
    for i, t in enumerate(prem.terms()):
        if isinstance(t, Constant):
            consts[i] = t.id
        else:
            var_pos[i] = t.name  # type: ignore[union-attr]
    probes = {i: b.col(v) for i, v in var_pos.items() if b.has(v)}
    if probes:
        li, s, p, o = scan_probe(idx, consts, probes)
        base = b.gather(li)
    else:
        s, p, o = scan_unit(idx, consts)
        cand_cols: Dict[str, torch.Tensor] = {}
        cols_all = (s, p, o)
        mask = torch.ones(s.numel(), dtype=torch.bool, device=device)
        for i, name in var_pos.items():
            if name in cand_cols:
                mask &= cand_cols[name] == cols_all[i]
            else:
                cand_cols[name] = cols_all[i]
        cand = Bindings(cand_cols, s.numel(), device).select(mask) \
            if not bool(mask.all()) else Bindings(cand_cols, s.numel(), device)
        from ..engine.executor import join_bindings
        return join_bindings(b, cand)
    # bind remaining (non-probed) vars; enforce repeated vars
    cols_all = (s, p, o)
    out_cols = dict(base.cols)
    mask = torch.ones(s.numel(), dtype=torch.bool, device=device)
    seen = set(probes.keys())
    bound_new: Dict[str, torch.Tensor] = {}
    for i, name in var_pos.items():
        if i in seen:
            continue
        if name in out_cols or name in bound_new:
            prev = bound_new.get(name, out_cols.get(name))
            mask &= prev == cols_all[i]
        else:
            bound_new[name] = cols_all[i]
    out_cols.update(bound_new)
    res = Bindings(out_cols, s.numel(), device)
    if not bool(mask.all()):
        res = res.select(mask)
    return res


def _apply_negative(b: Bindings, neg: List[TriplePattern], idx: GraphIndex,
                    device) -> Bindings:
    """NAF: drop bindings for which a negative premise matches a fact."""
    for prem in neg:
        if b.is_empty():
            return b
        consts: Dict[int, int] = {}
        probes: Dict[int, torch.Tensor] = {}
        ok = True
        for i, t in enumerate(prem.terms()):
            if isinstance(t, Constant):
                consts[i] = t.id
            elif isinstance(t, Variable) and b.has(t.name):
                probes[i] = b.col(t.name)
            else:
                ok = False  # unbound var in negation: matches any fact
        if not ok:
            s, _, _ = scan_unit(idx, consts)
            if s.numel() > 0:
                return Bindings.empty(device, b.variables)
            continue
        if probes:
            li, _, _, _ = scan_probe(idx, consts, probes)
            hit = torch.zeros(b.n, dtype=torch.bool, device=device)
            if li.numel():
                hit[li] = True
            b = b.select(~hit)
        else:
            s, _, _ = scan_unit(idx, consts)
            if s.numel() > 0:
                return Bindings.empty(device, b.variables)
    return b


def _instantiate(rule: Rule, b: Bindings, device
                 ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Conclusion instantiation (ref materialisation.rs:36
    replace_variables_with_bound_values) — columnar."""
    outs = [[], [], []]
    n = b.n
    for concl in rule.conclusion:
        cols = []
        ok = True
        for t in concl.terms():
            if isinstance(t, Constant):
                cols.append(torch.full((n,), t.id, dtype=torch.int32, device=device))
            elif isinstance(t, Variable):
                if not b.has(t.name):
                    ok = False
                    break
                cols.append(b.col(t.name))
            else:
                ok = False
                break
        if not ok:
            continue
        mask = (cols[0] != UNBOUND) & (cols[1] != UNBOUND) & (cols[2] != UNBOUND)
        if not bool(mask.all()):
            cols = [c[mask] for c in cols]
        outs[0].append(cols[0])
        outs[1].append(cols[1])
        outs[2].append(cols[2])
    if not outs[0]:
        e = torch.empty(0, dtype=torch.int32, device=device)
        return e, e.clone(), e.clone()
    return torch.cat(outs[0]), torch.cat(outs[1]), torch.cat(outs[2])


def _eval_filters(rule: Rule, b: Bindings, db) -> Bindings:
    for f in rule.filters:
        if b.is_empty():
            return b
        mask = f.eval_mask(b, db)
        b = b.select(mask)
    return b


def infer_round(
    rules: List[Rule],
    facts: FactStore,
    delta_s, delta_p, delta_o,
    db,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """One semi-naive round: for each rule and each premise position i,
    premise_i x delta joined with all facts for the others
    (ref semi_naive.rs:17-86)."""
    device = facts.device
    idx = facts.index()
    news = [[], [], []]
    for rule in rules:
        np_ = len(rule.premise)
        for i in range(np_):
            b = _match_premise_against_delta(
                rule.premise[i], delta_s, delta_p, delta_o, device)
            if b is None or b.is_empty():
                continue
            ok = True
            for j in range(np_):
                if j == i:
                    continue
                b = _join_premise_all_facts(b, rule.premise[j], idx, device)
                if b.is_empty():
                    ok = False
                    break
            if not ok:
                continue
            b = _eval_filters(rule, b, db)
            if b.is_empty():
                continue
            b = _apply_negative(b, rule.negative_premise, idx, device)
            if b.is_empty():
                continue
            s, p, o = _instantiate(rule, b, device)
            if s.numel():
                news[0].append(s)
                news[1].append(p)
                news[2].append(o)
    e = torch.empty(0, dtype=torch.int32, device=device)
    if not news[0]:
        return e, e.clone(), e.clone()
    s = torch.cat(news[0])
    p = torch.cat(news[1])
    o = torch.cat(news[2])
    # dedup within the round, then against known facts
    s, p, o = unique_rows([s, p, o])
    known = facts.sorted_unique_rows()
    if known[0].numel():
        hit = membership_mask([s, p, o], known)
        keep = ~hit
        s, p, o = s[keep], p[keep], o[keep]
    return s, p, o


def infer_fixpoint(rules: List[Rule], facts: FactStore, db,
                   semi_naive: bool = True, max_rounds: int = 10_000) -> int:
    """Loop until no new facts (ref infer_generic.rs:27).  Returns the
    number of derived facts."""
    device = facts.device
    total_new = 0
    if semi_naive:
        ds, dp, do_ = facts.s, facts.p, facts.o
    for _ in range(max_rounds):
        if semi_naive:
            s, p, o = infer_round(rules, facts, ds, dp, do_, db)
        else:
            s, p, o = infer_round(rules, facts, facts.s, facts.p, facts.o, db)
        if s.numel() == 0:
            break
        facts.add_columns(s, p, o)
        total_new += s.numel()
        if semi_naive:
            ds, dp, do_ = s, p, o
    return total_new
