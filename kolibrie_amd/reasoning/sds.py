"""Cross-window SDS+ reasoning: naive vs incremental with per-fact expiry.

Ref parity: datalog/src/cross_window_sds.rs (WindowedTriple with
event_time, per-window width, translate_sds_to_datalog -> (Triple, expiry)
:82), cross_window_naive.rs:20 (naive_sds_plus — full recompute over alive
facts) and cross_window_incremental.rs:26 (incremental_sds_plus —
SdsWithExpiry propagating per-fact expiry through the fixpoint via the
min/max ExpirationProvenance semiring, provenance.rs:460).

A derived fact lives while ALL its premises live: expiry(derived) =
min(expiry(premises)); under alternative derivations the max survives
(best proof) — exactly the Expiration semiring (plus = max, times = min).

Window contents are bounded, so this fixpoint runs on the host; the
unbounded base-store reasoning stays on the K6 columnar path.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Set, Tuple

from ..storage.terms import Constant, TriplePattern, Variable
from .rule import Rule

Triple = Tuple[int, int, int]


@dataclass(frozen=True)
class WindowedTriple:
    window_iri: str
    triple: Triple
    event_time: int


class Sds:
    """The streaming data snapshot: windowed triples with expiry."""

    def __init__(self):
        self.entries: List[Tuple[WindowedTriple, float]] = []

    def add(self, wt: WindowedTriple, width: int):
        self.entries.append((wt, wt.event_time + width))

    def alive_facts(self, ts: int) -> Dict[Triple, float]:
        """triple -> max expiry among alive occurrences."""
        out: Dict[Triple, float] = {}
        for wt, expiry in self.entries:
            if expiry > ts:
                prev = out.get(wt.triple)
                out[wt.triple] = expiry if prev is None else max(prev, expiry)
        return out

    def all_facts_with_expiry(self) -> Dict[Triple, float]:
        out: Dict[Triple, float] = {}
        for wt, expiry in self.entries:
            prev = out.get(wt.triple)
            out[wt.triple] = expiry if prev is None else max(prev, expiry)
        return out


def _i32(x: int) -> int:
    x &= 0xFFFFFFFF
    return x - 0x1_0000_0000 if x >= 0x8000_0000 else x


def _match(pattern: TriplePattern, fact: Triple,
           binding: Dict[str, int]) -> Optional[Dict[str, int]]:
    b = dict(binding)
    for term, val in zip(pattern.terms(), fact):
        v32 = _i32(val)
        if isinstance(term, Constant):
            if term.id != v32:
                return None
        elif isinstance(term, Variable):
            if term.name in b:
                if b[term.name] != v32:
                    return None
            else:
                b[term.name] = v32
        else:
            return None
    return b


def _instantiate(concl: TriplePattern, b: Dict[str, int]) -> Optional[Triple]:
    out = []
    for term in concl.terms():
        if isinstance(term, Constant):
            out.append(term.id & 0xFFFFFFFF)
        elif isinstance(term, Variable):
            if term.name not in b:
                return None
            out.append(b[term.name] & 0xFFFFFFFF)
        else:
            return None
    return tuple(out)  # type: ignore


def _expiry_fixpoint(facts: Dict[Triple, float], rules: List[Rule]
                     ) -> Dict[Triple, float]:
    """Semi-naive fixpoint propagating expiries (Expiration semiring:
    derived = min over premises, best derivation = max).  Tag-improved
    facts re-enter the delta (ref provenance_semi_naive.rs:185-197)."""
    known = dict(facts)
    delta = dict(facts)
    while delta:
        new_delta: Dict[Triple, float] = {}
        for rule in rules:
            np_ = len(rule.premise)
            for i in range(np_):
                # premise i matched against delta, the rest against known
                for dfact, dexp in delta.items():
                    b0 = _match(rule.premise[i], dfact, {})
                    if b0 is None:
                        continue
                    stack = [(b0, dexp, 0)]
                    while stack:
                        b, exp, j = stack.pop()
                        if j == np_:
                            for concl in rule.conclusion:
                                t = _instantiate(concl, b)
                                if t is None:
                                    continue
                                prev = known.get(t)
                                if prev is None or exp > prev:
                                    known[t] = exp
                                    nd = new_delta.get(t)
                                    new_delta[t] = exp if nd is None else max(nd, exp)
                            continue
                        if j == i:
                            stack.append((b, exp, j + 1))
                            continue
                        for f2, e2 in known.items():
                            b2 = _match(rule.premise[j], f2, b)
                            if b2 is not None:
                                stack.append((b2, min(exp, e2), j + 1))
        delta = new_delta
    return known


def naive_sds_plus(sds: Sds, rules: List[Rule], db, ts: int) -> Set[Triple]:
    """Full recompute over the facts alive at `ts`
    (ref cross_window_naive.rs:20)."""
    alive = sds.alive_facts(ts)
    known = _expiry_fixpoint(alive, rules)
    return {t for t, exp in known.items() if exp > ts}


def incremental_sds_plus(sds: Sds, rules: List[Rule], db, ts: int
                         ) -> Set[Triple]:
    """Expiry-propagating materialisation filtered at `ts`
    (ref cross_window_incremental.rs:26): the fixpoint runs over ALL facts
    with their expiries; liveness at ts is a filter on the derived expiry —
    equivalent to the naive recompute (tested)."""
    cache = getattr(sds, "_inc_cache", None)
    key = len(sds.entries)
    if cache is None or cache[0] != key:
        all_facts = sds.all_facts_with_expiry()
        known = _expiry_fixpoint(all_facts, rules)
        sds._inc_cache = (key, known)  # type: ignore[attr-defined]
    else:
        known = cache[1]
    return {t for t, exp in known.items() if exp > ts}
