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
    """Growing columnar fact set with an amortized-rebuild sorted index.

    The index covers a `base` prefix; facts added since the last rebuild
    live in a small unsorted `tail` that joins brute-force.  A rebuild
    (full sort) happens only when the tail outgrows 25% of the base, so a
    fixpoint of R rounds costs O(log R) sorts instead of R."""

    REBUILD_FRACTION = 0.25
    REBUILD_MIN = 16_384

    def __init__(self, device):
        self.device = torch.device(device)
        e = torch.empty(0, dtype=torch.int32, device=self.device)
        self.s, self.p, self.o = e, e.clone(), e.clone()
        self._index: Optional[GraphIndex] = None
        self._base_n = 0

    @property
    def n(self) -> int:
        return self.s.numel()

    def index(self) -> GraphIndex:
        """Sorted index over the base prefix (pair with tail_columns())."""
        if self._index is None or (
                self.n - self._base_n
                > max(self.REBUILD_MIN, self._base_n * self.REBUILD_FRACTION)):
            self._index = GraphIndex.from_columns(
                self.s, self.p, self.o, device=self.device, dedup=False)
            self._base_n = self.n
        return self._index

    def tail_columns(self):
        """Facts newer than the current index build."""
        return (self.s[self._base_n:], self.p[self._base_n:],
                self.o[self._base_n:])

    def add_columns(self, s, p, o) -> int:
        """Add new facts (must already be deduped against self); returns the
        number added."""
        if s.numel() == 0:
            return 0
        self.s = torch.cat([self.s, s])
        self.p = torch.cat([self.p, p])
        self.o = torch.cat([self.o, o])
        return s.numel()

    def set_columns(self, s, p, o):
        self.s, self.p, self.o = s, p, o
        self._index = None
        self._base_n = 0

    def sorted_unique_rows(self):
        return unique_rows([self.s, self.p, self.o])

    def remove(self, s: int, p: int, o: int) -> bool:
        """Remove one fact (repair application); True if it was present."""
        keep = ~((self.s == s) & (self.p == p) & (self.o == o))
        if bool(keep.all()):
            return False
        self.set_columns(self.s[keep], self.p[keep], self.o[keep])
        return True


def membership_in_index(idx: GraphIndex, s, p, o) -> torch.Tensor:
    """Vectorized membership of (s,p,o) rows in a sorted index — exact
    (s,p) searchsorted plus an in-range manual binary search on o.  No
    per-round sort of the known set (the old membership_mask lexsorted
    everything every round)."""
    from ..engine.tensor_utils import pack2
    from ..storage.dataset import SPO
    key12, z = idx.orders[SPO]
    n = s.numel()
    dev = s.device
    if n == 0 or idx.n == 0:
        return torch.zeros(n, dtype=torch.bool, device=dev)
    k = pack2(s, p)
    lo = torch.searchsorted(key12, k, side="left")
    hi = torch.searchsorted(key12, k, side="right")
    l, h = lo.clone(), hi.clone()
    zmax = z.numel() - 1
    while True:
        active = l < h
        if not bool(active.any()):
            break
        mid = (l + h) >> 1
        zm = z[torch.clamp(mid, max=zmax)]
        less = zm < o
        go_r = active & less
        go_l = active & ~less
        l = torch.where(go_r, mid + 1, l)
        h = torch.where(go_l, mid, h)
    in_range = l < hi
    zl = z[torch.clamp(l, max=zmax)]
    return in_range & (zl == o)


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
    b: Bindings, prem: TriplePattern, idx: GraphIndex, device,
    tail=None,
) -> Bindings:
    """Join current bindings with a premise matched against ALL facts
    (ref rules.rs:167 join_premise_with_hash_join -> K1/K2).  `tail` is
    the store's unsorted post-rebuild suffix, joined brute-force."""
    base_res = _join_premise_base(b, prem, idx, device)
    if tail is None or tail[0].numel() == 0:
        return base_res
    tail_cand = _match_premise_against_delta(prem, tail[0], tail[1], tail[2],
                                             device)
    if tail_cand is None or tail_cand.is_empty():
        return base_res
    from ..engine.executor import join_bindings
    tail_res = join_bindings(b, tail_cand)
    return Bindings.concat([base_res, tail_res], device)


def _join_premise_base(
    b: Bindings, prem: TriplePattern, idx: GraphIndex, device
) -> Bindings:
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
                    device, tail=None) -> Bindings:
    """NAF: drop bindings for which a negative premise matches a fact."""
    for prem in neg:
        if tail is not None and tail[0].numel() and not b.is_empty():
            tail_cand = _match_premise_against_delta(
                prem, tail[0], tail[1], tail[2], device)
            if tail_cand is not None and not tail_cand.is_empty():
                from ..engine.executor import join_bindings
                marked = b.with_col(
                    "__row", torch.arange(b.n, dtype=torch.int32,
                                          device=device))
                hit_rows = join_bindings(marked, tail_cand)
                hit = torch.zeros(b.n, dtype=torch.bool, device=device)
                if not hit_rows.is_empty() and hit_rows.has("__row"):
                    hit[hit_rows.col("__row").to(torch.long)] = True
                b = b.select(~hit)
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
    tail = facts.tail_columns()
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
                b = _join_premise_all_facts(b, rule.premise[j], idx, device,
                                            tail)
                if b.is_empty():
                    ok = False
                    break
            if not ok:
                continue
            b = _eval_filters(rule, b, db)
            if b.is_empty():
                continue
            b = _apply_negative(b, rule.negative_premise, idx, device, tail)
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
    # dedup within the round, then against known facts: searchsorted
    # membership against the sorted base + mask against the small tail
    s, p, o = unique_rows([s, p, o])
    keep = ~membership_in_index(idx, s, p, o)
    if tail[0].numel():
        tail_sorted = unique_rows(list(tail))
        keep &= ~membership_mask([s, p, o], tail_sorted)
    s, p, o = s[keep], p[keep], o[keep]
    return s, p, o


def infer_fixpoint(rules: List[Rule], facts: FactStore, db,
                   semi_naive: bool = True, max_rounds: int = 10_000) -> int:
    """Loop until no new facts (ref infer_generic.rs:27).  Returns the
    number of derived facts."""
    device = facts.device
    # Small working sets are launch-latency-bound on the columnar path
    # (deep-taxonomy shape: thousands of tiny rounds).  Order of
    # preference: (1) K6 persistent single-workgroup device kernel — the
    # whole fixpoint in ONE launch, rounds are __syncthreads boundaries
    # (eligible 1-2-premise const-predicate programs); (2) host hash
    # fixpoint, kept as the fallback optimization for small programs the
    # kernel's rule language doesn't cover; (3) columnar device rounds.
    from .host_fixpoint import HOST_PATH_MAX_FACTS, infer_fixpoint_host
    if semi_naive and device.type == "cuda" \
            and facts.n <= HOST_PATH_MAX_FACTS:
        from .device_fixpoint import try_device_fixpoint
        res = try_device_fixpoint(rules, facts, db)
        if res is not None:
            return res
    if semi_naive and facts.n <= HOST_PATH_MAX_FACTS:
        tuples = list(zip(facts.s.cpu().tolist(), facts.p.cpu().tolist(),
                          facts.o.cpu().tolist()))
        derived = infer_fixpoint_host(rules, tuples, db)
        if derived:
            import numpy as np
            arr = np.asarray(derived, dtype=np.int64).astype(np.int32)
            t = torch.from_numpy(arr).to(device)
            facts.add_columns(t[:, 0].contiguous(), t[:, 1].contiguous(),
                              t[:, 2].contiguous())
        return len(derived)
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
