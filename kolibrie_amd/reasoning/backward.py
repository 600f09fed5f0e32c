"""Backward chaining: SLD-style goal resolution with unification.

Ref parity: datalog/src/reasoning/backward_chaining.rs:7-206 —
unify_terms (:27, incl. quoted triples), goal resolution against facts and
rule heads with standardized-apart rule variables, returning the bindings
that prove the goal.
"""
from __future__ import annotations

from typing import Dict, Iterator, List, Optional, Tuple

from ..storage.terms import Constant, QuotedTriplePattern, TriplePattern, Variable
from .rule import Rule

Triple = Tuple[int, int, int]
# substitution: var name -> Constant | Variable (chains resolved by walk)
Subst = Dict[str, object]


def _i32(x: int) -> int:
    x &= 0xFFFFFFFF
    return x - 0x1_0000_0000 if x >= 0x8000_0000 else x


def walk(t, s: Subst):
    """Resolve a term through the substitution chain."""
    while isinstance(t, Variable) and t.name in s:
        t = s[t.name]
    return t


def unify_terms(a, b, s: Optional[Subst], qt_store=None) -> Optional[Subst]:
    """Unify two terms under substitution `s` (ref backward_chaining.rs:27).
    Handles Variable/Constant and quoted-triple patterns vs quoted ids."""
    if s is None:
        return None
    a = walk(a, s)
    b = walk(b, s)
    if isinstance(a, Variable):
        if isinstance(b, Variable) and b.name == a.name:
            return s
        out = dict(s)
        out[a.name] = b
        return out
    if isinstance(b, Variable):
        out = dict(s)
        out[b.name] = a
        return out
    if isinstance(a, Constant) and isinstance(b, Constant):
        return s if a.id == b.id else None
    if isinstance(a, QuotedTriplePattern) and isinstance(b, Constant):
        if qt_store is None:
            return None
        t = qt_store.decode(b.id & 0xFFFFFFFF)
        if t is None:
            return None
        for sub_a, val in zip((a.s, a.p, a.o), t):
            s = unify_terms(sub_a, Constant(_i32(val)), s, qt_store)
            if s is None:
                return None
        return s
    if isinstance(b, QuotedTriplePattern):
        return unify_terms(b, a, s, qt_store)
    if isinstance(a, QuotedTriplePattern) and isinstance(b, QuotedTriplePattern):
        for x, y in zip((a.s, a.p, a.o), (b.s, b.p, b.o)):
            s = unify_terms(x, y, s, qt_store)
            if s is None:
                return None
        return s
    return None


def _unify_pattern(a: TriplePattern, b: TriplePattern, s: Subst,
                   qt_store=None) -> Optional[Subst]:
    for x, y in zip(a.terms(), b.terms()):
        s = unify_terms(x, y, s, qt_store)
        if s is None:
            return None
    return s


def _fact_pattern(fact: Triple) -> TriplePattern:
    return TriplePattern(*(Constant(_i32(v)) for v in fact))


def _rename(rule: Rule, tag: int) -> Rule:
    """Standardize apart rule variables per resolution step."""
    def ren(p: TriplePattern) -> TriplePattern:
        def r(t):
            if isinstance(t, Variable):
                return Variable(f"{t.name}${tag}")
            if isinstance(t, QuotedTriplePattern):
                return QuotedTriplePattern(r(t.s), r(t.p), r(t.o))
            return t
        return TriplePattern(r(p.s), r(p.p), r(p.o))
    return Rule(
        premise=[ren(p) for p in rule.premise],
        negative_premise=[ren(p) for p in rule.negative_premise],
        conclusion=[ren(c) for c in rule.conclusion],
        name=rule.name,
    )


def backward_chain(goal: TriplePattern, reasoner, max_depth: int = 24
                   ) -> List[Dict[str, int]]:
    """Prove `goal` against the reasoner's facts and rules; returns the
    goal-variable bindings that satisfy it (ref :150)."""
    reasoner._flush()
    facts = sorted(reasoner.all_fact_tuples())
    rules = reasoner.rules
    qt_store = getattr(reasoner.db, "quoted_triples", None)
    counter = [0]

    def solve(goals: List[TriplePattern], s: Subst, depth: int
              ) -> Iterator[Subst]:
        if not goals:
            yield s
            return
        if depth > max_depth:
            return
        first, rest = goals[0], goals[1:]
        for fact in facts:
            s2 = _unify_pattern(first, _fact_pattern(fact), s, qt_store)
            if s2 is not None:
                yield from solve(rest, s2, depth)
        for rule in rules:
            counter[0] += 1
            r = _rename(rule, counter[0])
            for concl in r.conclusion:
                s2 = _unify_pattern(first, concl, s, qt_store)
                if s2 is not None:
                    yield from solve(list(r.premise) + rest, s2, depth + 1)

    out: List[Dict[str, int]] = []
    goal_vars = goal.variables()
    dedup = set()
    for s in solve([goal], {}, 0):
        proj = {}
        ok = True
        for v in goal_vars:
            t = walk(Variable(v), s)
            if isinstance(t, Constant):
                proj[v] = t.id & 0xFFFFFFFF
            else:
                ok = False
        if ok or not goal_vars:
            key = tuple(sorted(proj.items()))
            if key not in dedup:
                dedup.add(key)
                out.append(proj)
    return out
