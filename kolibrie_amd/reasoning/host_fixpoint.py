"""Host-side hash semi-naive fixpoint — the small-working-set fast path.

The columnar K6 fixpoint pays ~40 device ops per round; on chain-shaped
workloads (deep taxonomy: 10 000 rounds of 1-fact deltas) that fixed cost
dominates.  Below `HOST_PATH_MAX_FACTS` the engine switches to this
dict-indexed implementation (the moral equivalent of the reference's
HashMap joins, semi_naive.rs) — microseconds per tiny round — while large
working sets stay on the device path.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict, List, Optional, Set, Tuple

from ..storage.terms import Constant, TriplePattern, Variable
from .rule import Rule

Triple = Tuple[int, int, int]

HOST_PATH_MAX_FACTS = 200_000


def _i32(x: int) -> int:
    x &= 0xFFFFFFFF
    return x - 0x1_0000_0000 if x >= 0x8000_0000 else x


class _FactIndex:
    """Adjacency maps: (pos_key) -> facts, pos in {s, o, so}."""

    __slots__ = ("by_sp", "by_op", "by_p", "all")

    def __init__(self):
        self.by_sp: Dict[Tuple[int, int], List[Triple]] = defaultdict(list)
        self.by_op: Dict[Tuple[int, int], List[Triple]] = defaultdict(list)
        self.by_p: Dict[int, List[Triple]] = defaultdict(list)
        self.all: Set[Triple] = set()

    def add(self, t: Triple) -> bool:
        if t in self.all:
            return False
        self.all.add(t)
        s, p, o = t
        self.by_sp[(s, p)].append(t)
        self.by_op[(o, p)].append(t)
        self.by_p[p].append(t)
        return True

    def candidates(self, prem: TriplePattern, b: Dict[str, int]) -> List[Triple]:
        def val(term) -> Optional[int]:
            if isinstance(term, Constant):
                return term.id
            if isinstance(term, Variable) and term.name in b:
                return b[term.name]
            return None

        s, p, o = val(prem.s), val(prem.p), val(prem.o)
        if p is not None:
            if s is not None:
                return self.by_sp.get((s, p), [])
            if o is not None:
                return self.by_op.get((o, p), [])
            return self.by_p.get(p, [])
        return list(self.all)


def _match(prem: TriplePattern, fact: Triple, b: Dict[str, int]
           ) -> Optional[Dict[str, int]]:
    out = None
    for term, val in zip(prem.terms(), fact):
        if isinstance(term, Constant):
            if term.id != val:
                return None
        elif isinstance(term, Variable):
            cur = (out or b).get(term.name)
            if cur is None:
                if out is None:
                    out = dict(b)
                out[term.name] = val
            elif cur != val:
                return None
        else:
            return None
    return out if out is not None else dict(b)


def _eval_filters(rule: Rule, b: Dict[str, int], db) -> bool:
    if not rule.filters:
        return True
    import torch
    from ..engine.bindings import Bindings
    row = Bindings({k: torch.tensor([v], dtype=torch.int32)
                    for k, v in b.items()}, 1, "cpu")
    return all(bool(f.eval_mask(row, db).item()) for f in rule.filters)


def infer_fixpoint_host(rules: List[Rule], fact_tuples: List[Triple], db
                        ) -> List[Triple]:
    """Semi-naive over host dict indexes; returns newly derived facts
    (i32 component tuples)."""
    idx = _FactIndex()
    for t in fact_tuples:
        idx.add(t)
    delta: List[Triple] = list(idx.all)
    derived: List[Triple] = []
    while delta:
        new: List[Triple] = []
        for rule in rules:
            np_ = len(rule.premise)
            for i in range(np_):
                prem_i = rule.premise[i]
                for dfact in delta:
                    b0 = _match(prem_i, dfact, {})
                    if b0 is None:
                        continue
                    stack = [(b0, 0)]
                    while stack:
                        b, j = stack.pop()
                        if j == np_:
                            if not _eval_filters(rule, b, db):
                                continue
                            if any(_ground(neg, b) in idx.all
                                   for neg in rule.negative_premise
                                   if _ground(neg, b) is not None):
                                continue
                            for concl in rule.conclusion:
                                t = _ground(concl, b)
                                if t is not None and idx.add(t):
                                    new.append(t)
                                    derived.append(t)
                            continue
                        if j == i:
                            stack.append((b, j + 1))
                            continue
                        for f2 in idx.candidates(rule.premise[j], b):
                            b2 = _match(rule.premise[j], f2, b)
                            if b2 is not None:
                                stack.append((b2, j + 1))
        delta = new
    return derived


def _ground(p: TriplePattern, b: Dict[str, int]) -> Optional[Triple]:
    out = []
    for term in p.terms():
        if isinstance(term, Constant):
            out.append(term.id)
        elif isinstance(term, Variable):
            v = b.get(term.name)
            if v is None:
                return None
            out.append(v)
        else:
            return None
    return (out[0], out[1], out[2])
