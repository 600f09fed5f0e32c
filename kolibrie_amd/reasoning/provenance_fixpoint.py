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
    tags: Dict[Triple, object] = dict(seeds)
    positive_rules = [r for r in rules if not r.negative_premise]
    naf_rules = [r for r in rules if r.negative_premise]

    def _close(a, b) -> bool:
        if isinstance(a, float) and isinstance(b, float):
            return abs(a - b) <= 1e-12
        return a == b

    def _contributions(active_rules, cur: Dict[Triple, object],
                       use_naf: bool) -> Dict[Triple, object]:
        """One naive evaluation pass over `cur`: ⊕ of every distinct rule
        derivation.  Recomputing from a SNAPSHOT each round (Jacobi) is
        what makes non-idempotent semirings (noisy-or AddMult) correct —
        the reference's delta re-entry (provenance_semi_naive.rs:185-197)
        re-⊕s the same derivation on tag improvement, which only works
        for idempotent ⊕ (MinMax/Boolean); the oracle must not
        double-count."""
        contrib: Dict[Triple, object] = {}
        items = list(cur.items())
        for rule in active_rules:
            np_ = len(rule.premise)

            def rec(j, b, tag):
                if j == np_:
                    if db is not None and not _eval_filters(rule, b, db):
                        return
                    final_tag = tag
                    if use_naf and rule.negative_premise:
                        final_tag = _apply_naf(rule, b, cur, semiring,
                                               final_tag)
                        if final_tag is None:
                            return
                    for concl in rule.conclusion:
                        t = _instantiate(concl, b)
                        if t is None:
                            continue
                        prev = contrib.get(t)
                        contrib[t] = final_tag if prev is None else \
                            semiring.plus(prev, final_tag)
                    return
                for f2, t2 in items:
                    b2 = _match(rule.premise[j], f2, b)
                    if b2 is not None:
                        rec(j + 1, b2, semiring.times(tag, t2))

            rec(0, {}, semiring.one())
        return contrib

    # stratum 0: positive rules to fixpoint (Jacobi iteration)
    for _ in range(max_rounds):
        contrib = _contributions(positive_rules, tags, use_naf=False)
        new_tags: Dict[Triple, object] = dict(seeds)
        for t, c in contrib.items():
            base = new_tags.get(t)
            new_tags[t] = c if base is None else semiring.plus(base, c)
        if (set(new_tags) == set(tags)
                and all(_close(new_tags[t], tags[t]) for t in tags)):
            break
        tags = new_tags
    # stratum 1: one NAF pass (ref run_negative_stratum_pass:297-389)
    if naf_rules:
        contrib = _contributions(naf_rules, tags, use_naf=True)
        for t, c in contrib.items():
            prev = tags.get(t)
            tags[t] = c if prev is None else semiring.plus(prev, c)
    return tags


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
