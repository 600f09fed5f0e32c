"""SPARQL UPDATE execution (ref: execute_query.rs:500-796 — WHERE evaluated
once, deletes before inserts, per-solution blank-node allocation :603-629,
RDF-legality checks :727-796)."""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from ..parsing.ast import GGP, GBgp, QuadData, UpdateOperation
from ..storage.dataset import DEFAULT_GRAPH
from ..storage.terms import Constant, UNBOUND, Variable
from .bindings import Bindings
from .executor import DatasetView, ExecutionContext, ExecutionEngine

_BNODE_COUNTER = [0]


def _is_literal_surface(t: str) -> bool:
    return t.strip()[:1] in "\"'"


def _is_var(t: str) -> bool:
    return t.strip()[:1] in "?$"


def execute_update(op: UpdateOperation, db, prefixes: Dict[str, str]):
    if op.kind == "insert_data":
        for q in op.quads:
            _check_data_legality(q)
            _insert_quad(db, q, prefixes)
        db.store.commit_all()
        return
    if op.kind == "delete_data":
        for q in op.quads:
            _check_data_legality(q, allow_bnode=False)
            _delete_quad(db, q, prefixes)
        db.store.commit_all()
        return
    if op.kind == "delete_where":
        where = GBgp([type("TP", (), {})() for _ in ()])  # placeholder
        from ..parsing.ast import TriplePatternAst
        pats = [TriplePatternAst(q.s, q.p, q.o) for q in op.delete_templates]
        # GRAPH-scoped templates evaluate inside their graph
        rows = _eval_where_for_templates(db, prefixes, op.delete_templates)
        _apply_templates(db, prefixes, op.delete_templates, rows, delete=True)
        db.store.commit_all()
        return
    if op.kind == "modify":
        rows = _eval_where(db, prefixes, op.where)
        _apply_templates(db, prefixes, op.delete_templates, rows, delete=True)
        _apply_templates(db, prefixes, op.insert_templates, rows, delete=False)
        db.store.commit_all()
        return
    if op.kind in ("clear", "drop", "create"):
        _graph_management(db, op, prefixes)
        return
    raise ValueError(f"unknown update op {op.kind}")


def _check_data_legality(q: QuadData, allow_bnode: bool = True):
    """Ground-data legality (ref execute_query.rs:727-796): no variables in
    DATA blocks; literals are illegal in subject/predicate position."""
    for pos, t in (("subject", q.s), ("predicate", q.p), ("object", q.o)):
        if _is_var(t):
            raise ValueError(f"variable in {pos} of a DATA block")
    if _is_literal_surface(q.s):
        raise ValueError("literal in subject position")
    if _is_literal_surface(q.p):
        raise ValueError("literal in predicate position")
    if not allow_bnode:
        if q.s.startswith("_:") or q.o.startswith("_:"):
            raise ValueError("blank node in DELETE DATA")


def _gid_of(db, g: Optional[str], prefixes) -> int:
    if g is None:
        return DEFAULT_GRAPH
    return db.dictionary.encode(db.resolve_lexical(g, prefixes))


def _insert_quad(db, q: QuadData, prefixes):
    db.store.insert_quad(
        _gid_of(db, q.g, prefixes),
        db.encode_term_star(q.s, prefixes),
        db.encode_term_star(q.p, prefixes),
        db.encode_term_star(q.o, prefixes),
    )


def _delete_quad(db, q: QuadData, prefixes):
    ids = []
    for t in (q.s, q.p, q.o):
        lex = db.resolve_lexical(t, prefixes)
        i = db.dictionary.lookup(lex) if not t.startswith("<<") else db.encode_term_star(t, prefixes)
        if i is None:
            return
        ids.append(i)
    db.store.delete_quad(_gid_of(db, q.g, prefixes), *ids)


def _eval_where(db, prefixes, where: Optional[GGP]) -> Bindings:
    from ..plan.lower import build_logical_plan
    from ..plan.optimizer import Streamertail
    if where is None:
        return Bindings.unit(db.device)
    stats = db.get_or_build_stats()
    plan = build_logical_plan(where, db, prefixes)
    phys = Streamertail(stats).find_best_plan(plan)
    ctx = ExecutionContext(db, DatasetView())
    return ExecutionEngine(ctx).execute(phys, Bindings.unit(db.device))


def _eval_where_for_templates(db, prefixes, templates: List[QuadData]) -> Bindings:
    from ..parsing.ast import GBgp, GGraph, GJoin, GUnit, TriplePatternAst
    node: GGP = GUnit()
    cur_default: List[TriplePatternAst] = []
    for q in templates:
        pat = TriplePatternAst(q.s, q.p, q.o)
        if q.g is None:
            cur_default.append(pat)
        else:
            sub = GGraph(q.g, GBgp([pat]))
            node = sub if isinstance(node, GUnit) else GJoin(node, sub)
    if cur_default:
        bgp = GBgp(cur_default)
        node = bgp if isinstance(node, GUnit) else GJoin(node, bgp)
    return _eval_where(db, prefixes, node)


def _apply_templates(db, prefixes, templates: List[QuadData], rows: Bindings,
                     delete: bool):
    if not templates:
        return
    n = rows.n
    if n == 0:
        return
    host_cols = {v: (rows.col(v).to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
                 for v in rows.variables}
    unbound_mask = {v: (rows.col(v) == UNBOUND).cpu().tolist()
                    for v in rows.variables}
    for q in templates:
        # GRAPH ?g templates take the graph id from each solution's binding
        # (ref sparql_graph_test.rs graph_variables_flow_from_where_into_
        # delete_and_insert_templates)
        g_ids = None
        if q.g is not None and _is_var(q.g.strip()):
            gname = q.g.strip()[1:]
            if gname not in host_cols:
                continue
            g_ids = host_cols[gname]
            gid = None
        else:
            gid = _gid_of(db, q.g, prefixes)
        bnode_map: Dict[str, List[int]] = {}
        term_ids = []
        ok_rows = [True] * n
        for t in (q.s, q.p, q.o):
            t = t.strip()
            if _is_var(t):
                name = t[1:]
                if name not in host_cols:
                    ok_rows = [False] * n
                    term_ids.append(None)
                    continue
                for i in range(n):
                    if unbound_mask[name][i]:
                        ok_rows[i] = False
                term_ids.append(host_cols[name])
            elif t.startswith("_:") and not delete:
                # fresh blank node per solution (ref :603-629)
                ids = []
                for i in range(n):
                    # fresh label: skip ids already interned (persisted
                    # lexical collisions must not alias two bnodes — ref
                    # update_blank_node_allocation_skips_persisted_
                    # lexical_collisions)
                    while True:
                        _BNODE_COUNTER[0] += 1
                        label = f"_:upd{_BNODE_COUNTER[0]}"
                        if db.dictionary.lookup(label) is None:
                            break
                    ids.append(db.dictionary.encode(label))
                term_ids.append(ids)
            else:
                tid = db.encode_term_star(t, prefixes)
                term_ids.append([tid] * n)
        if any(c is None for c in term_ids):
            continue
        # template legality: skip solutions producing literal subjects
        lit_subject = _is_literal_surface(q.s)
        if lit_subject:
            continue
        for i in range(n):
            if not ok_rows[i]:
                continue
            row_gid = g_ids[i] if g_ids is not None else gid
            if g_ids is not None and unbound_mask.get(q.g.strip()[1:],
                                                      [False] * n)[i]:
                continue
            s, p, o = term_ids[0][i], term_ids[1][i], term_ids[2][i]
            if delete:
                db.store.delete_quad(row_gid, s, p, o)
            else:
                db.store.insert_quad(row_gid, s, p, o)


def _graph_management(db, op: UpdateOperation, prefixes):
    if op.graph == "ALL":
        for g in list(db.store.graphs):
            db.store.clear_graph(g)
        if op.kind == "drop":
            for g in db.store.named_graph_ids():
                db.store.drop_graph(g)
        return
    if op.graph == "NAMED":
        for g in db.store.named_graph_ids():
            if op.kind == "clear":
                db.store.clear_graph(g)
            else:
                db.store.drop_graph(g)
        return
    if op.graph is None:
        if op.kind in ("clear", "drop"):
            db.store.clear_graph(DEFAULT_GRAPH)
        return
    gid = _gid_of(db, op.graph, prefixes)
    if op.kind == "clear":
        if gid not in db.store.graphs and not op.silent:
            raise ValueError("graph does not exist")
        db.store.clear_graph(gid)
    elif op.kind == "create":
        if gid in db.store.catalog and not op.silent:
            raise ValueError("graph already exists")
        db.store.create_graph(gid)
    elif op.kind == "drop":
        existed = db.store.drop_graph(gid)
        if not existed and not op.silent:
            raise ValueError("graph does not exist")
