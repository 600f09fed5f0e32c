"""AST -> logical plan lowering (ref: streamertail_optimizer/utils.rs:402-577
build_logical_plan_from_group, :192 compile_term).

Dictionary ENCODE writes happen here (constants are interned at compile
time), so execution is pure-int32.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..engine.filters import CompiledBind, CompiledExpr
from ..parsing.ast import (
    GBgp, GBind, GFilter, GGP, GGraph, GJoin, GMinus, GOptional, GSubQuery,
    GUnion,
    GUnit, GValues, GWindowBlock, SelectQuery, TriplePatternAst,
)
from ..storage.terms import Constant, QuotedTriplePattern, TriplePattern, Variable
from .logical import (
    GraphScope, LBind, LJoin, LScan, LSelection,
    LSubquery, LUnion, LUnit, LValues, LogicalOp,
)


def _to_i32(x: int) -> int:
    x &= 0xFFFFFFFF
    return x - 0x1_0000_0000 if x >= 0x8000_0000 else x


def compile_term(term_str: str, prefixes: Dict[str, str], db):
    """Surface term string -> Variable / Constant / QuotedTriplePattern."""
    t = term_str.strip()
    if t.startswith("?") or t.startswith("$"):
        return Variable(t[1:])
    if t.startswith("<<") and t.endswith(">>"):
        from ..storage.database import split_quoted_triple_content
        s_s, p_s, o_s = split_quoted_triple_content(t[2:-2].strip())
        s_t = compile_term(s_s, prefixes, db)
        p_t = compile_term(p_s, prefixes, db)
        o_t = compile_term(o_s, prefixes, db)
        if all(isinstance(x, Constant) for x in (s_t, p_t, o_t)):
            qid = db.quoted_triples.encode(
                s_t.id & 0xFFFFFFFF, p_t.id & 0xFFFFFFFF, o_t.id & 0xFFFFFFFF
            )
            return Constant(_to_i32(qid))
        return QuotedTriplePattern(s_t, p_t, o_t)
    return Constant(_to_i32(db.dictionary.encode(db.resolve_lexical(t, prefixes))))


def compile_triple_pattern(p: TriplePatternAst, prefixes, db) -> TriplePattern:
    return TriplePattern(
        compile_term(p.s, prefixes, db),
        compile_term(p.p, prefixes, db),
        compile_term(p.o, prefixes, db),
    )


def compile_graph_term(g: str, prefixes, db) -> GraphScope:
    g = g.strip()
    if g.startswith("?") or g.startswith("$"):
        return ("var", g[1:])
    gid = db.dictionary.encode(db.resolve_lexical(g, prefixes))
    return ("const", _to_i32(gid))


@dataclass
class CompiledSubquery:
    """A sub-SELECT retained with its own modifiers (ref utils.rs subquery
    spec; finalized inside the engine, engine.rs:785-905)."""
    select: SelectQuery
    plan: LogicalOp
    prefixes: Dict[str, str] = field(default_factory=dict)
    physical: object = None  # filled by the optimizer


def build_logical_plan(
    g: GGP, db, prefixes: Dict[str, str], scope: GraphScope = None
) -> LogicalOp:
    if isinstance(g, GUnit):
        return LUnit()
    if isinstance(g, GBgp):
        node: LogicalOp = LUnit()
        for pat in g.patterns:
            cp = compile_triple_pattern(pat, prefixes, db)
            pat_scope = scope
            if getattr(pat, "path_mod", None):
                # p+ / p* closure steps scan a materialized closure index
                # (built lazily by the executor from the device fixpoint)
                if not isinstance(cp.p, Constant):
                    raise ValueError(
                        "property-path closure requires a constant predicate")
                if scope is not None and scope[0] != "const":
                    raise ValueError(
                        "property-path closure under GRAPH ?var is "
                        "unsupported")
                gid = None if scope is None else scope[1] & 0xFFFFFFFF
                pat_scope = ("closure", cp.p.id & 0xFFFFFFFF,
                             pat.path_mod == "*", gid)
            scan = LScan(cp, pat_scope)
            node = scan if isinstance(node, LUnit) else LJoin(node, scan)
        return node
    if isinstance(g, GJoin):
        return LJoin(
            build_logical_plan(g.left, db, prefixes, scope),
            build_logical_plan(g.right, db, prefixes, scope),
        )
    if isinstance(g, GUnion):
        return LUnion(
            build_logical_plan(g.left, db, prefixes, scope),
            build_logical_plan(g.right, db, prefixes, scope),
        )
    if isinstance(g, GGraph):
        inner_scope = compile_graph_term(g.graph, prefixes, db)
        return build_logical_plan(g.inner, db, prefixes, inner_scope)
    if isinstance(g, GFilter):
        return LSelection(
            CompiledExpr(g.expr, db, prefixes),
            build_logical_plan(g.inner, db, prefixes, scope),
        )
    if isinstance(g, GBind):
        return LBind(
            CompiledBind(g.expr, db, prefixes),
            g.var,
            build_logical_plan(g.inner, db, prefixes, scope),
        )
    if isinstance(g, GValues):
        rows: List[List[Optional[int]]] = []
        for row in g.rows:
            crow: List[Optional[int]] = []
            for cell in row:
                if cell is None:
                    crow.append(None)
                else:
                    term = compile_term(cell, prefixes, db)
                    crow.append(term.id if isinstance(term, Constant) else None)
            rows.append(crow)
        return LValues(list(g.variables), rows,
                       build_logical_plan(g.inner, db, prefixes, scope))
    if isinstance(g, GSubQuery):
        sub_plan = build_logical_plan(g.select.where, db, prefixes, scope)
        return LSubquery(
            CompiledSubquery(g.select, sub_plan, dict(prefixes)),
            build_logical_plan(g.inner, db, prefixes, scope),
        )
    if isinstance(g, GWindowBlock):
        # outside the RSP runtime a WINDOW block lowers to its inner pattern
        # (the RSP builder splits blocks per window before lowering)
        return build_logical_plan(g.inner, db, prefixes, scope)
    if isinstance(g, GMinus):
        from .logical import LMinus
        return LMinus(
            build_logical_plan(g.left, db, prefixes, scope),
            build_logical_plan(g.right, db, prefixes, scope),
        )
    if isinstance(g, GOptional):
        from .logical import LLeftJoin
        return LLeftJoin(
            build_logical_plan(g.left, db, prefixes, scope),
            build_logical_plan(g.right, db, prefixes, scope),
        )
    raise ValueError(f"cannot lower {type(g).__name__}")
