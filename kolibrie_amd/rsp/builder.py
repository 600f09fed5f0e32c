"""RSPBuilder — fluent construction of RSPEngine from RSP-QL text.

Ref parity: kolibrie/src/rsp/builder.rs (381 LoC): add_rsp_ql_query /
add_triples / add_rules / add_reasoning_rules / add_sparql_rules /
add_cross_window_rules / set_sync_policy / set_operation_mode /
set_query_execution_mode / add_r2r / add_consumer; build() parses the
REGISTER clause, creates one window + optimized physical plan per WINDOW
block (:212-324), and wires the engine.
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

from ..parsing.ast import (
    GGP, GBgp, GFilter, GJoin, GUnit, GWindowBlock, Projection, SelectQuery,
    SyncPolicy,
)
from ..parsing.sparql import parse_combined_query
from .engine import (
    CrossWindowReasoningMode, OperationMode, QueryExecutionMode, RSPEngine,
    normalize_iri,
)


class RSPBuilder:
    def __init__(self, device: str = "cpu"):
        self.device = device
        self._rsp_ql: Optional[str] = None
        self._triples: List[Tuple[str, str, str]] = []
        self._static_nt: List[str] = []
        self._rules_text: List[str] = []
        self._sparql_rules: List[str] = []
        self._cross_window_rules: List[str] = []
        self._sync_policy: Optional[SyncPolicy] = None
        self._operation_mode = OperationMode.SINGLE_THREAD
        self._query_mode = QueryExecutionMode.VOLCANO
        self._cross_window_mode: Optional[str] = None
        self._consumers: List[Callable] = []
        self._r2r = None
        self._hybrid_config = None

    # ------------------------------------------------------------- fluent
    def add_rsp_ql_query(self, text: str) -> "RSPBuilder":
        self._rsp_ql = text
        return self

    def add_triples(self, triples) -> "RSPBuilder":
        self._triples.extend(triples)
        return self

    def add_static_ntriples(self, text: str) -> "RSPBuilder":
        self._static_nt.append(text)
        return self

    def add_rules(self, text: str) -> "RSPBuilder":
        self._rules_text.append(text)
        return self

    add_reasoning_rules = add_rules

    def add_sparql_rules(self, text: str) -> "RSPBuilder":
        self._sparql_rules.append(text)
        return self

    def add_cross_window_rules(self, text: str) -> "RSPBuilder":
        self._cross_window_rules.append(text)
        if self._cross_window_mode is None:
            self._cross_window_mode = CrossWindowReasoningMode.INCREMENTAL
        return self

    def set_sync_policy(self, kind: str, timeout_ms: Optional[int] = None
                        ) -> "RSPBuilder":
        self._sync_policy = SyncPolicy(kind, timeout_ms)
        return self

    def set_operation_mode(self, mode: str) -> "RSPBuilder":
        self._operation_mode = mode
        return self

    def set_query_execution_mode(self, mode: str) -> "RSPBuilder":
        self._query_mode = mode
        return self

    def set_cross_window_mode(self, mode: str) -> "RSPBuilder":
        self._cross_window_mode = mode
        return self

    def set_hybrid_config(self, cfg) -> "RSPBuilder":
        self._hybrid_config = cfg
        return self

    def add_r2r(self, r2r) -> "RSPBuilder":
        self._r2r = r2r
        return self

    def add_consumer(self, fn: Callable) -> "RSPBuilder":
        self._consumers.append(fn)
        return self

    # -------------------------------------------------------------- build
    def build(self) -> RSPEngine:
        if self._rsp_ql is None:
            raise ValueError("RSPBuilder requires an RSP-QL query")
        cq = parse_combined_query(self._rsp_ql)
        reg = cq.register
        if reg is None:
            raise ValueError("RSP-QL query must contain a REGISTER clause")
        engine = RSPEngine(
            register=reg, device=self.device,
            operation_mode=self._operation_mode,
            sync_policy=self._sync_policy
            or (reg.windows[0].policy if reg.windows and reg.windows[0].policy
                else SyncPolicy("Wait")),
            cross_window_mode=self._cross_window_mode,
            hybrid_config=self._hybrid_config,
        )
        if self._r2r is not None:
            engine.store = self._r2r
        db = engine.store.db
        prefixes = dict(db.prefixes)
        prefixes.update(cq.prefixes)

        # base data and rules
        for t in self._triples:
            engine.store.add(t)
        for nt in self._static_nt:
            engine.load_static_ntriples(nt)
        for rt in self._rules_text + self._sparql_rules:
            engine.store.load_rules(rt)
        for cr in cq.rules:
            from ..reasoning.rule import convert_combined_rule
            engine.store.add_rule(convert_combined_rule(cr, db, prefixes))
        # PROB(provenance=hybrid, ...) on any rule turns on hybrid window
        # evaluation (ref rsp_engine_test.rs hybrid_rsp_rule: the annotation,
        # not an explicit config, selects the hybrid path)
        if engine.store.hybrid_config is None:
            for r in engine.store.rules:
                pa = getattr(r, "prob", None)
                if pa is not None and pa.provenance == "hybrid":
                    from ..reasoning.hybrid import HybridConfig
                    cfg = HybridConfig()
                    if pa.threshold is not None:
                        cfg.threshold = pa.threshold
                    cfg.policy = pa.extra.get("threshold_policy", "Explicit")
                    if "band_epsilon" in pa.extra:
                        cfg.band_epsilon = float(pa.extra["band_epsilon"])
                    for k, attr in (("k_initial", "k_initial"),
                                    ("k_max", "k_max"),
                                    ("k_growth", "k_growth"),
                                    ("node_budget", "sdd_node_cap")):
                        if k in pa.extra:
                            setattr(cfg, attr, int(pa.extra[k]))
                    if "topk_budget_ms" in pa.extra:
                        cfg.topk_budget_ms = float(pa.extra["topk_budget_ms"])
                    if "sdd_budget_ms" in pa.extra:
                        cfg.sdd_budget_ms = float(pa.extra["sdd_budget_ms"])
                    cfg.validate()
                    engine.store.hybrid_config = cfg
                    break
        for cw in self._cross_window_rules:
            from ..reasoning.n3_rules import parse_n3_rules_for_sds
            engine.cross_window_rules.extend(
                parse_n3_rules_for_sds(cw, db, engine._sds_window_widths))

        # window blocks -> per-window patterns
        blocks = _collect_window_blocks(reg.select.where)
        per_window_ggp: Dict[str, GGP] = {}
        for iri, inner in blocks:
            key = normalize_iri(iri)
            if key in per_window_ggp:
                per_window_ggp[key] = GJoin(per_window_ggp[key], inner)
            else:
                per_window_ggp[key] = inner
        engine.projection = reg.select

        for wc in reg.windows:
            wiri = normalize_iri(wc.window_iri)
            inner = per_window_ggp.get(wiri)
            plan = None
            plan_vars: List[str] = []
            if inner is not None:
                agg_select = reg.select if (
                    len(reg.windows) == 1
                    and any(p.aggregate for p in reg.select.variables)
                ) else None
                plan, plan_vars = _build_window_plan(inner, db, prefixes,
                                                     agg_select)
            engine.add_window(
                wiri, wc.stream_iri, wc.spec.width, wc.spec.slide or wc.spec.width,
                report=wc.spec.report, tick=wc.spec.tick,
                plan=plan, plan_vars=plan_vars)
            if self._sync_policy is None and wc.policy is not None:
                engine.sync_policy = wc.policy
        for fn in self._consumers:
            engine.add_consumer(fn)
        return engine


def _collect_window_blocks(g: GGP) -> List[Tuple[str, GGP]]:
    out: List[Tuple[str, GGP]] = []

    def rec(x: GGP):
        if isinstance(x, GWindowBlock):
            out.append((x.window_iri, x.inner))
            return
        if isinstance(x, GJoin):
            rec(x.left)
            rec(x.right)
        elif isinstance(x, GFilter):
            rec(x.inner)
        elif hasattr(x, "inner"):
            rec(x.inner)

    rec(g)
    return out


def _build_window_plan(inner: GGP, db, prefixes, agg_select=None):
    """Build an optimized physical plan projecting all window variables;
    for a single-window aggregate query the register's own SELECT (with
    aggregates) finalizes on device (ref builder.rs:279-324)."""
    from ..plan.lower import build_logical_plan
    from ..plan.optimizer import Streamertail, annotate_needed
    logical = build_logical_plan(inner, db, prefixes)
    stats = db.get_or_build_stats()
    physical = Streamertail(stats).find_best_plan(logical)
    from ..plan.optimizer import _logical_out_vars
    if agg_select is not None:
        from ..engine.query import _top_needed
        sel = SelectQuery(
            variables=list(agg_select.variables),
            group_by=list(agg_select.group_by),
            distinct=agg_select.distinct, where=inner)
        annotate_needed(physical, _top_needed(sel))
        plan_vars = [p.output_name() for p in sel.variables]
        return (sel, physical), plan_vars
    annotate_needed(physical, None)
    plan_vars = sorted(_logical_out_vars(logical))
    sel = SelectQuery(variables=[Projection(var=v) for v in plan_vars],
                      where=inner)
    return (sel, physical), plan_vars
