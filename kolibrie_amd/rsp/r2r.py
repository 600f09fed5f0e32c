"""R2R: relation-to-relation operators.

Ref parity: kolibrie/src/rsp/r2r.rs:17-30 (R2ROperator trait) and
rsp/simple_r2r.rs:30-220 (SimpleR2R: SparqlDatabase + reasoning rules +
optional hybrid config; materialize() evicts previously derived triples,
runs semi-naive inference and re-adds derived facts).

MI355X redesign: the reference rebuilds a fresh Reasoner and copies every
DB triple per window firing (simple_r2r.rs:154-158); here the window store
is already device-resident columns and inference runs the K6 columnar
fixpoint directly over them.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Dict, List, Tuple

from ..storage.database import SparqlDatabase
from ..storage.dataset import DEFAULT_GRAPH


class R2ROperator(ABC):
    """load_triples/load_rules/add/remove/materialize/execute_query
    (ref r2r.rs:21-30)."""

    @abstractmethod
    def load_triples(self, triples, ts: int = 0):
        ...

    @abstractmethod
    def load_rules(self, rules_text: str):
        ...

    @abstractmethod
    def add(self, triple, ts: int):
        ...

    @abstractmethod
    def remove(self, triple):
        ...

    @abstractmethod
    def materialize(self) -> List[Tuple[int, int, int]]:
        ...

    @abstractmethod
    def execute_query(self, plan) -> List[Tuple]:
        ...


class SimpleR2R(R2ROperator):
    """Default R2R over a SparqlDatabase with datalog rules."""

    def __init__(self, device: str = "cpu", hybrid_config=None):
        self.db = SparqlDatabase(device=device)
        self.rules = []              # reasoning Rule objects
        self.hybrid_config = hybrid_config
        self.seed_snapshot = None    # live hybrid seed snapshot
        self._derived: List[Tuple[int, int, int]] = []
        self.latest_hybrid_results: Dict = {}

    # ---------------------------------------------------------------- load
    def load_triples(self, triples, ts: int = 0):
        """triples: iterable of (s,p,o) string tuples or encoded id tuples."""
        for t in triples:
            self.add(t, ts)

    def load_rules(self, rules_text: str):
        """Parse N3 `{ premise } => { conclusion }` rules or RULE syntax
        (ref simple_r2r.rs:113-133)."""
        text = rules_text.strip()
        if not text:
            return
        if "=>" in text and "CONSTRUCT" not in text.upper():
            from ..reasoning.n3_rules import parse_n3_rules
            self.rules.extend(parse_n3_rules(text, self.db))
        else:
            from ..parsing.sparql import parse_combined_query
            from ..reasoning.rule import convert_combined_rule
            cq = parse_combined_query(text)
            pfx = dict(self.db.prefixes)
            pfx.update(cq.prefixes)
            for cr in cq.rules:
                self.rules.append(convert_combined_rule(cr, self.db, pfx))

    def add_rule(self, rule):
        self.rules.append(rule)

    # ----------------------------------------------------------- mutation
    def _encode(self, triple) -> Tuple[int, int, int]:
        s, p, o = triple
        if isinstance(s, str):
            return (self.db.encode_term_star(s), self.db.encode_term_star(p),
                    self.db.encode_term_star(o))
        return (s & 0xFFFFFFFF, p & 0xFFFFFFFF, o & 0xFFFFFFFF)

    def add(self, triple, ts: int = 0):
        s, p, o = self._encode(triple)
        self.db.store.insert_quad(DEFAULT_GRAPH, s, p, o)

    def remove(self, triple):
        s, p, o = self._encode(triple)
        self.db.store.delete_quad(DEFAULT_GRAPH, s, p, o)

    # -------------------------------------------------------- materialize
    def materialize(self) -> List[Tuple[int, int, int]]:
        """Evict previous derivations, run the fixpoint over the current
        window content, re-add derived facts (ref simple_r2r.rs:143-199).
        Returns the newly derived (s,p,o) id triples."""
        # evict previous firing's derived triples
        for (s, p, o) in self._derived:
            self.db.store.delete_quad(DEFAULT_GRAPH, s, p, o)
        self._derived = []
        if not self.rules:
            return []
        from ..reasoning.reasoner import Reasoner
        from ..reasoning.seminaive import infer_fixpoint, FactStore
        idx = self.db.store.graph_index(DEFAULT_GRAPH)
        s, p, o = idx.columns()
        facts = FactStore(self.db.device)
        facts.add_columns(s, p, o)
        base_n = facts.n
        if self.hybrid_config is not None and self.seed_snapshot is not None:
            from ..reasoning.hybrid import evaluate_hybrid_rules
            derived, results = evaluate_hybrid_rules(
                self.rules, facts, self.db, self.hybrid_config,
                self.seed_snapshot)
            self.latest_hybrid_results = results
        else:
            infer_fixpoint(self.rules, facts, self.db, semi_naive=True)
            derived = None
        import torch
        new_s = facts.s[base_n:]
        new_p = facts.p[base_n:]
        new_o = facts.o[base_n:]
        su = (new_s.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        pu = (new_p.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        ou = (new_o.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        out = list(zip(su, pu, ou))
        if derived is not None:
            out = derived
        for (a, b, c) in out:
            self.db.store.insert_quad(DEFAULT_GRAPH, a, b, c)
        self._derived = out
        return out

    # -------------------------------------------------------------- query
    def execute_query(self, plan) -> List[Tuple[str, ...]]:
        """Run a prepared physical plan over the store, returning sorted
        decoded rows (ref simple_r2r.rs:201-214)."""
        from ..engine.finalize import decode_rows
        select, final = self.execute_query_bindings(plan)
        return sorted(tuple(r) for r in decode_rows(select, final, self.db))

    def execute_query_bindings(self, plan):
        """Columnar variant: returns (select, finalized Bindings) WITHOUT
        decoding — the engine's single-window emission keeps results as
        device columns through R2S (K10 rows_diff) and decodes only the
        emitted Δ."""
        from ..engine.bindings import Bindings
        from ..engine.executor import (DatasetView, ExecutionContext,
                                       ExecutionEngine)
        from ..engine.finalize import finalize_select_bindings
        select, physical = plan
        ctx = ExecutionContext(self.db, DatasetView())
        rows = ExecutionEngine(ctx).execute(physical,
                                            Bindings.unit(self.db.device))
        return select, finalize_select_bindings(select, rows, self.db)

    def execute_sparql(self, sparql: str) -> List[List[str]]:
        return self.db.query(sparql)
