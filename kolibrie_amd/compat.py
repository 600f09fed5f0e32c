"""Reference-compatible Python API surface.

The reference exposes PyO3 bindings (python/src/lib.rs module `kolibrie`:
PySparqlDatabase, PyKnowledgeGraph, the fluent query builder); this
framework's host runtime IS Python, so the bindings collapse to aliases
over the native classes — the public behavior contract is the same:

    PySparqlDatabase  -> SparqlDatabase (add_triple, parse_turtle,
                         load_file with format sniffing, exec_query,
                         update, query builder)
    PyKnowledgeGraph  -> Reasoner (abox, rules, infer naive/semi-naive,
                         backward chaining, constraints, repairs)
"""
from .engine.query_builder import QueryBuilder  # noqa: F401
from .reasoning.reasoner import Reasoner  # noqa: F401
from .storage.database import SparqlDatabase as PySparqlDatabase  # noqa: F401


class PyKnowledgeGraph(Reasoner):
    """Reference-shaped PyKnowledgeGraph: the infer_* entry points
    return the NEWLY derived triples as decoded (s, p, o) string tuples
    (py_knowledge_graph.rs:174-211), not the engine-native new-fact
    count.  Everything else is the Reasoner surface unchanged."""

    def _snapshot(self):
        self._flush()
        return set(self.all_fact_tuples())

    def _decoded_diff(self, before):
        dec = self.dictionary.decode
        out = []
        for (s, p, o) in sorted(set(self.all_fact_tuples()) - before):
            out.append((dec(s) or "", dec(p) or "", dec(o) or ""))
        return out

    def infer_new_facts(self):
        before = self._snapshot()
        super().infer_new_facts()
        return self._decoded_diff(before)

    def infer_new_facts_semi_naive(self):
        before = self._snapshot()
        super().infer_new_facts_semi_naive()
        return self._decoded_diff(before)

    infer_new_facts_semi_naive_parallel = infer_new_facts_semi_naive

    def infer_new_facts_semi_naive_with_repairs(self):
        before = self._snapshot()
        super().infer_new_facts_semi_naive_with_repairs()
        return self._decoded_diff(before)


# ---- Datalog rule-construction surface (py_knowledge_graph.rs:25-110) ----
# The reference registers Rule / TriplePattern / Term / FilterCondition
# classes for building rules from Python.  Term.Variable / Term.Constant
# map onto this framework's Variable / Constant terms; FilterCondition
# evaluates with the reference's semantics (datalog rules.rs:133-165):
# variable-vs-variable compares dictionary IDs for =/!=, otherwise the
# bound value parses as f64 (0.0 fallback) against the literal.

from .reasoning.rule import Rule  # noqa: F401,E402
from .storage.terms import (  # noqa: F401,E402
    Constant, TriplePattern, Variable,
)


class _TermNamespace:
    """`Term.Variable("x")` / `Term.Constant(42)` constructors."""

    Variable = Variable
    Constant = Constant


Term = _TermNamespace


class FilterCondition:
    """Rule filter with the reference's numeric/id semantics; implements
    the engine's eval_mask protocol so it drops straight into
    Rule.filters."""

    __slots__ = ("variable", "operator", "value")

    def __init__(self, variable: str, operator: str, value: str):
        self.variable = variable
        self.operator = operator
        self.value = value

    def __repr__(self):
        return (f"FilterCondition({self.variable!r}, {self.operator!r}, "
                f"{self.value!r})")

    def eval_mask(self, b, db):
        import torch
        var = self.variable.lstrip("?")
        val = self.value.lstrip("?") if isinstance(self.value, str) else ""
        n = b.n
        dev = next(iter(b.cols.values())).device if b.cols else "cpu"
        if not b.has(var):
            return torch.ones(n, dtype=torch.bool, device=dev)
        lhs = b.cols[var]
        if b.has(val):
            rhs = b.cols[val]
            if self.operator == "=":
                return lhs == rhs
            if self.operator == "!=":
                return lhs != rhs
            return torch.ones(n, dtype=torch.bool, device=dev)
        vc = db.value_column()
        if vc.device != lhs.device:
            vc = vc.to(lhs.device)
        nums = vc[(lhs.to(torch.int64) & 0xFFFFFFFF).clamp(
            max=vc.numel() - 1)]
        try:
            target = float(self.value)
        except (TypeError, ValueError):
            target = 0.0
        op = self.operator
        if op == ">":
            return nums > target
        if op == "<":
            return nums < target
        if op == ">=":
            return nums >= target
        if op == "<=":
            return nums <= target
        if op == "=":
            return (nums - target).abs() <= torch.finfo(torch.float64).eps
        if op == "!=":
            return (nums - target).abs() > torch.finfo(torch.float64).eps
        return torch.ones(n, dtype=torch.bool, device=dev)


def Rule_from_parts(premise, filters, conclusion):
    """Reference PyRule ctor shape: Rule(premise, filters, conclusion)."""
    return Rule(premise=list(premise), filters=list(filters),
                conclusion=list(conclusion))

# Enum/name exports matching the reference module registrations
# (py_query_builder.rs:814-826): StreamingQuery is the builder itself
# (as_stream returns the same fluent object), SortDirection maps to
# asc()/desc(), and the RSP enums re-export the engine's constants.
from .rsp.s2r import ReportStrategy, Tick  # noqa: F401,E402
from .rsp.r2s import StreamOperator  # noqa: F401,E402

StreamingQuery = QueryBuilder


class SortDirection:
    Ascending = "Ascending"
    Descending = "Descending"


class PeriodicReportStrategy:
    """Periodic report with a period (reference PyPeriodicReportStrategy:
    holds the period used by with_periodic_report)."""

    __slots__ = ("period",)

    def __init__(self, period: int):
        self.period = period
