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
from .reasoning.reasoner import Reasoner as PyKnowledgeGraph  # noqa: F401
from .storage.database import SparqlDatabase as PySparqlDatabase  # noqa: F401
