"""Provenance-tagged semi-naive materialisation.

Ref parity: datalog/src/materialisation/provenance_semi_naive.rs (390 LoC):
  - every fact carries a semiring tag; seeds sorted by triple for
    deterministic TopK variable ids (:220-228);
  - tag-IMPROVED existing facts re-enter the delta (`delta_improved`,
    :28,185-197) — the fixpoint triggers on tag change, not only new facts;
  - stratified NAF: positive fixpoint (stratum 0), then a single negative
    pass (:297-389): absent negated fact -> one(), present -> negate(tag),
    ⊗-combined into the rule tag.

Scalar-tag semirings (MinMax / AddMult / Expiration) can ride the K6
columnar fixpoint as an f32 tag column; this host implementation is the
general/oracle path covering structured tags (TopK proofs, DNF-WMC).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..storage.terms import Constant, TriplePattern, Variable
from .provenance import Provenance
from .rule import Rule

Triple = Tuple[int, int, int]


def _i32(x: int) -> int:
    x &= 0xFFFFFFFF
    return x - 0x1_0000_0000 if x >= 0x8000_0000 else x


def _match(pattern: TriplePattern, fact: Triple, binding: Dict[str, int]
           ) -> Optional[Dict[str, int]]:
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
    return (out[0], out[1], out[2])


def _eval_filters(rule: Rule, b: Dict[str, int], db) -> bool:
    if not rule.filters:
        return True
    from ..engine.bindings import Bindings
    import torch
    row = Bindings(
        {k: torch.tensor([v], dtype=torch.int32) for k, v in b.items()},
        1, "cpu")
    for f in rule.filters:
        if not bool(f.eval_mask(row, db).item()):
            return False
    return True


def infer_with_provenance(
    rules: List[Rule],
    seeds: Dict[Triple, object],
    semiring: Provenance,
    db=None,
    max_rounds: int = 10_000,
) -> Dict[Triple, object]:
    """Positive tagged fixpoint + one stratified NAF pass.

    `seeds`: triple -> initial tag.  Returns triple -> final tag for all
    facts (seeds included, possibly tag-improved).
    """
    known: Dict[Triple, object] = dict(seeds)
    delta: Dict[Triple, object] = dict(seeds)
    positive_rules = [r for r in rules if not r.negative_premise]
    naf_rules = [r for r in rules if r.negative_premise]

    def run_round(active_rules, use_naf_stage: bool):
        nonlocal known, delta
        rounds = 0
        while delta and rounds < max_rounds:
            rounds += 1
            new_delta: Dict[Triple, object] = {}
            for rule in active_rules:
                np_ = len(rule.premise)
                for i in range(np_):
                    for dfact, dtag in delta.items():
                        b0 = _match(rule.premise[i], dfact, {})
                        if b0 is None:
                            continue
                        stack = [(b0, dtag, 0)]
                        while stack:
                            b, tag, j = stack.pop()
                            if j == np_:
                                if db is not None and not _eval_filters(rule, b, db):
                                    continue
                                final_tag = tag
                                if use_naf_stage and rule.negative_premise:
                                    final_tag = _apply_naf(
                                        rule, b, known, semiring, final_tag)
                                    if final_tag is None:
                                        continue
                                for concl in rule.conclusion:
                                    t = _instantiate(concl, b)
                                    if t is None:
                                        continue
                                    prev = known.get(t)
                                    if prev is None:
                                        known[t] = final_tag
                                        new_delta[t] = final_tag
                                    else:
                                        merged = semiring.plus(prev, final_tag)
                                        if merged != prev and semiring.better(
                                                merged, prev):
                                            # tag-improved fact re-enters Δ
                                            known[t] = merged
                                            new_delta[t] = merged
                                        elif merged != prev:
                                            known[t] = merged
                                continue
                            if j == i:
                                stack.append((b, tag, j + 1))
                                continue
                            for f2, t2 in known.items():
                                b2 = _match(rule.premise[j], f2, b)
                                if b2 is not None:
                                    stack.append(
                                        (b2, semiring.times(tag, t2), j + 1))
            delta = new_delta

    # stratum 0: positive rules to fixpoint
    run_round(positive_rules, use_naf_stage=False)
    # stratum 1: one NAF pass (ref run_negative_stratum_pass:297-389)
    if naf_rules:
        delta = dict(known)
        run_round(naf_rules, use_naf_stage=True)
    return known


def _apply_naf(rule: Rule, b: Dict[str, int], known: Dict[Triple, object],
               semiring: Provenance, tag):
    """⊗ in the negated premises' tags: absent -> one(), present ->
    negate(tag); unsupported negation kills the derivation."""
    for neg in rule.negative_premise:
        t = _instantiate(neg, b)
        if t is None:
            return None
        present = known.get(t)
        if present is None:
            factor = semiring.one()
        else:
            try:
                factor = semiring.negate(present)
            except NotImplementedError:
                return None
        tag = semiring.times(tag, factor)
    return tag
