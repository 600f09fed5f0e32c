"""Neural relation registry + materialization.

Ref parity: kolibrie/src/neural_relations.rs (908 LoC) — registry of
MODEL / NEURAL RELATION / TRAIN declarations (:63), execute_train_decl
(:250), materialize_neural_relation (:447): run the model over anchor
entities bound by the INPUT patterns and assert predicate triples (plus
probability companion facts) BEFORE query/rule evaluation (:539);
ML.PREDICT alias lowering (:553).
"""
from __future__ import annotations

from typing import Dict

import torch

from ..engine.bindings import Bindings
from .feature_loader import features_from_bindings

DEFAULT_THRESHOLD = 0.5


def _rows_for_patterns(patterns, db, prefixes) -> Bindings:
    from ..engine.executor import DatasetView, ExecutionContext, ExecutionEngine
    from ..parsing.ast import GBgp, TriplePatternAst
    from ..plan.lower import build_logical_plan
    from ..plan.optimizer import Streamertail, annotate_needed
    ggp = GBgp([TriplePatternAst(*p) if isinstance(p, tuple) else p
                for p in patterns])
    logical = build_logical_plan(ggp, db, prefixes)
    plan = Streamertail(db.get_or_build_stats()).find_best_plan(logical)
    annotate_needed(plan, None)
    ctx = ExecutionContext(db, DatasetView())
    return ExecutionEngine(ctx).execute(plan, Bindings.unit(db.device))


def materialize_neural_relation(name: str, entry: dict, db,
                                prefixes: Dict[str, str]):
    """Evaluate INPUT patterns, run the model, assert relation triples
    (ref neural_relations.rs:447-539)."""
    decl = entry["decl"]
    model_entry = db.neural_models.get(decl.model)
    model = model_entry.get("model") if model_entry else None
    if model is None:
        return 0
    feat_vars = [v for v in decl.options.get("features", "").split(",") if v]
    patterns = decl.inputs
    if not patterns:
        return 0
    rows = _rows_for_patterns(patterns, db, prefixes)
    if rows.is_empty():
        return 0
    x = features_from_bindings(rows, feat_vars, db)
    norm = (model_entry or {}).get("norm")
    if norm is not None:
        x = (x - norm[0]) / norm[1]
    proba = model.predict_proba(x)
    # anchor = the subject variable of the first input pattern
    anchor_term = patterns[0][0]
    anchor_var = anchor_term[1:] if anchor_term.startswith("?") else None
    if anchor_var is None or not rows.has(anchor_var):
        return 0
    pred_id = db.dictionary.encode(db.resolve_lexical(decl.name, prefixes))
    labels = (model_entry or {}).get("labels") or ["true"]
    # re-materialization first removes THIS relation's previous assertions
    # — never user facts with the same predicate nor other conclusions
    # (ref neural_relations.rs:488 remove_materialized_triples +
    # neural_materialized_triples registry; tests rerun_cleans_stale_
    # predictions / preserves_non_ml_conclusions)
    tracked = db.neural_materialized_triples.setdefault(pred_id, [])
    for (ts, tp, to) in tracked:
        db.store.delete_quad(0, ts, tp, to)
        db.probability_seeds.pop((ts, tp, to), None)
    tracked.clear()
    anchors = (rows.col(anchor_var).to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    n_asserted = 0
    if proba.dim() > 1 and proba.shape[1] > 1 and len(labels) > 1:
        # exclusive (multiclass) output: assert the argmax label per anchor
        # (ref NeuralOutputKind::Exclusive, neural_relations.rs:492-511)
        best = proba.argmax(dim=1).cpu().tolist()
        conf = proba.max(dim=1).values.detach().cpu().tolist()
        label_ids = [db.dictionary.encode(l) for l in labels]
        for a, bi, p in zip(anchors, best, conf):
            obj = label_ids[bi] if bi < len(label_ids) else label_ids[-1]
            db.store.insert_quad(0, a, pred_id, obj)
            db.probability_seeds[(a, pred_id, obj)] = p
            tracked.append((a, pred_id, obj))
            n_asserted += 1
        return n_asserted
    if proba.dim() > 1:
        proba = proba[:, -1]
    positive = db.dictionary.encode(labels[0])
    threshold = float(decl.options.get("threshold", DEFAULT_THRESHOLD))
    for a, p in zip(anchors, proba.detach().cpu().tolist()):
        if p >= threshold:
            db.store.insert_quad(0, a, pred_id, positive)
            db.probability_seeds[(a, pred_id, positive)] = p
            tracked.append((a, pred_id, positive))
            n_asserted += 1
    return n_asserted


def materialize_for_select(select, db, prefixes: Dict[str, str]):
    """If the query references a declared neural predicate, materialize it
    first (ref materialize_neural_relations_for_patterns)."""
    if not db.neural_relations:
        return
    from ..parsing.ast import GBgp, GFilter, GGP, GJoin
    pred_surfaces = set()

    def rec(g):
        if isinstance(g, GBgp):
            for p in g.patterns:
                pred_surfaces.add(db.resolve_lexical(p.p, prefixes))
        elif isinstance(g, GJoin):
            rec(g.left)
            rec(g.right)
        elif hasattr(g, "inner"):
            rec(g.inner)

    rec(select.where)
    for name, entry in db.neural_relations.items():
        if db.resolve_lexical(name, prefixes) in pred_surfaces:
            materialize_neural_relation(name, entry, db, prefixes)
