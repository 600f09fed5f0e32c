"""QueryEngine — thin in-memory convenience wrapper + explain
(ref: kolibrie/src/query_engine.rs:17-120)."""
from __future__ import annotations

from typing import List, Optional

from ..storage.database import SparqlDatabase


class QueryEngine:
    def __init__(self, device: str = "cpu", db: Optional[SparqlDatabase] = None):
        self.db = db if db is not None else SparqlDatabase(device=device)

    def add_triple(self, s: str, p: str, o: str):
        self.db.add_triple(s, p, o)

    def query(self, sparql: str) -> List[List[str]]:
        return self.db.query(sparql)

    def explain(self, sparql: str) -> str:
        """Render the optimized physical plan (ref query_engine.rs explain)."""
        from ..parsing.sparql import parse_combined_query
        from ..plan.lower import build_logical_plan
        from ..plan.optimizer import Streamertail, annotate_needed
        from ..plan.physical import (
            PBind, PBindJoin, PFilter, PHashJoin, PIndexScan, PMinus,
            PNestedLoopJoin, PStarJoin, PSubquery, PTableScan, PUnion,
            PUnit, PValues,
        )
        cq = parse_combined_query(sparql)
        if cq.select is None:
            return "(not a SELECT query)"
        prefixes = dict(self.db.prefixes)
        prefixes.update(cq.prefixes)
        logical = build_logical_plan(cq.select.where, self.db, prefixes)
        plan = Streamertail(self.db.get_or_build_stats()).find_best_plan(logical)
        from ..engine.query import _top_needed
        annotate_needed(plan, _top_needed(cq.select))
        lines: List[str] = []

        def fmt_pattern(p):
            def t(x):
                from ..storage.terms import Constant, Variable
                if isinstance(x, Variable):
                    return f"?{x.name}"
                if isinstance(x, Constant):
                    return self.db.decode_term(x.id & 0xFFFFFFFF) or f"#{x.id}"
                return "<<qt>>"
            return f"{t(p.s)} {t(p.p)} {t(p.o)}"

        def rec(op, indent):
            pad = "  " * indent
            name = type(op).__name__[1:]
            if isinstance(op, (PTableScan, PIndexScan)):
                hint = " [subject-sorted]" if getattr(op, "sort_hint", None) == 0 else ""
                lines.append(f"{pad}{name}({fmt_pattern(op.pattern)}){hint}")
            elif isinstance(op, PStarJoin):
                lines.append(f"{pad}StarJoin(?{op.join_var})")
                for p in op.patterns:
                    lines.append(f"{pad}  {fmt_pattern(p)}")
            elif isinstance(op, (PHashJoin, PBindJoin, PNestedLoopJoin,
                                 PUnion, PMinus)):
                lines.append(pad + name)
                rec(op.left, indent + 1)
                rec(op.right, indent + 1)
            elif isinstance(op, PFilter):
                lines.append(f"{pad}Filter")
                rec(op.input, indent + 1)
            elif hasattr(op, "input"):
                lines.append(pad + name)
                rec(op.input, indent + 1)
            else:
                lines.append(pad + name)

        rec(plan, 0)
        return "\n".join(lines)
