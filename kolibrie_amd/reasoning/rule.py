"""Datalog rule model (ref: shared/src/rule.rs:22-57 — premise,
negative_premise (NAF), filters, conclusion; check_rule_safety) and the
RuleIndex (shared/src/rule_index.rs:19-227 — 6 permutation indexes with
WILDCARD keys, 8-case candidate dispatch)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set

from ..storage.terms import Constant, TriplePattern, Variable

WILDCARD = 0xFFFFFFFF


@dataclass
class Rule:
    premise: List[TriplePattern] = field(default_factory=list)
    negative_premise: List[TriplePattern] = field(default_factory=list)
    filters: List[object] = field(default_factory=list)   # CompiledExpr
    conclusion: List[TriplePattern] = field(default_factory=list)
    name: str = ""
    prob: Optional[object] = None   # ProbAnnotation

    def head_variables(self) -> Set[str]:
        out: Set[str] = set()
        for c in self.conclusion:
            out.update(c.variables())
        return out

    def body_variables(self) -> Set[str]:
        out: Set[str] = set()
        for p in self.premise:
            out.update(p.variables())
        return out

    def check_safety(self) -> bool:
        """Every head/negative/filter variable must occur in a positive
        premise (ref rule.rs check_rule_safety)."""
        body = self.body_variables()
        if not self.head_variables() <= body:
            return False
        for np_ in self.negative_premise:
            if not set(np_.variables()) <= body:
                return False
        return True


def convert_combined_rule(cr, db, prefixes: Dict[str, str]) -> Rule:
    """CombinedRule AST -> Rule (ref parser.rs:3412 convert_combined_rule)."""
    from ..engine.filters import CompiledExpr
    from ..parsing.ast import GBgp, GFilter, GGP, GJoin, GUnit
    from ..plan.lower import compile_triple_pattern

    premises: List[TriplePattern] = []
    filters: List[object] = []

    def walk(g: GGP):
        if isinstance(g, GBgp):
            for p in g.patterns:
                premises.append(compile_triple_pattern(p, prefixes, db))
        elif isinstance(g, GJoin):
            walk(g.left)
            walk(g.right)
        elif isinstance(g, GFilter):
            filters.append(CompiledExpr(g.expr, db, prefixes))
            walk(g.inner)
        elif isinstance(g, GUnit):
            pass
        else:
            # window blocks etc: flatten inner patterns
            inner = getattr(g, "inner", None)
            if inner is not None:
                walk(inner)

    walk(cr.body)
    negated = [compile_triple_pattern(p, prefixes, db) for p in cr.negated]
    conclusions = [compile_triple_pattern(p, prefixes, db) for p in cr.conclusions]
    return Rule(
        premise=premises,
        negative_premise=negated,
        filters=filters,
        conclusion=conclusions,
        name=cr.name,
        prob=cr.prob,
    )


class RuleIndex:
    """Maps (predicate-ish keys) -> rule ids so a delta triple finds its
    candidate rules without scanning all rules (ref rule_index.rs)."""

    def __init__(self):
        self.rules: List[Rule] = []
        # per premise position: predicate id (or WILDCARD) -> rule indexes
        self.by_pred: Dict[int, Set[int]] = {}
        self.wildcard: Set[int] = set()

    def add_rule(self, rule: Rule) -> int:
        rid = len(self.rules)
        self.rules.append(rule)
        for prem in rule.premise:
            if isinstance(prem.p, Constant):
                self.by_pred.setdefault(prem.p.id & 0xFFFFFFFF, set()).add(rid)
            else:
                self.wildcard.add(rid)
        return rid

    def candidates_for_predicate(self, pid: int) -> Set[int]:
        out = set(self.wildcard)
        out |= self.by_pred.get(pid & 0xFFFFFFFF, set())
        return out

    def all_rules(self) -> List[Rule]:
        return self.rules
