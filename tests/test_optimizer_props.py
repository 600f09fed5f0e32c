"""Optimizer unit tests (mirrors streamertail_optimizer/optimizer.rs:907-1198:
star detection, greedy ordering properties, join-connectivity invariant,
hand-built stats fixtures as the mock backend)."""
import pytest

from kolibrie_amd.plan.cost import CostEstimator
from kolibrie_amd.plan.logical import LScan
from kolibrie_amd.plan.optimizer import Streamertail, _logical_out_vars
from kolibrie_amd.plan.physical import (
    PBindJoin, PHashJoin, PIndexScan, PNestedLoopJoin, PTableScan,
)
from kolibrie_amd.plan.stats import DatabaseStats
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def chain_stats() -> DatabaseStats:
    """Hand-built stats fixture (ref optimizer.rs:1063 chain_stats)."""
    st = DatabaseStats()
    st.total = 100_000
    st.pred_count = {1: 50_000, 2: 40_000, 3: 100}
    st.pred_distinct_subj = {1: 50_000, 2: 40_000, 3: 100}
    st.pred_distinct_obj = {1: 500, 2: 40_000, 3: 100}
    st.distinct_subjects = 50_000
    st.distinct_objects = 40_000
    return st


def _scan(s, p, o):
    def t(x):
        return Variable(x[1:]) if isinstance(x, str) and x.startswith("?") \
            else Constant(x)
    return LScan(TriplePattern(t(s), t(p), t(o)))


def _scans_of(plan):
    out = []

    def rec(op):
        if isinstance(op, (PIndexScan, PTableScan)):
            out.append(op)
        for attr in ("left", "right", "input"):
            child = getattr(op, attr, None)
            if child is not None:
                rec(child)

    rec(plan)
    return out


def test_greedy_picks_selective_anchor_first():
    st = Streamertail(chain_stats())
    scans = [_scan("?a", 1, "?b"), _scan("?b", 2, "?c"), _scan("?c", 3, "?d")]
    ordered = st._greedy_order(scans, set())
    # predicate 3 has cardinality 100 — the cheapest anchored seed
    assert ordered[0].pattern.p.id == 3


def test_greedy_join_connectivity_invariant():
    st = Streamertail(chain_stats())
    scans = [_scan("?a", 1, "?b"), _scan("?b", 2, "?c"), _scan("?c", 2, "?d")]
    ordered = st._greedy_order(scans, set())
    # every pick after the seed must share a variable with the bound set
    bound = set(ordered[0].pattern.variables())
    for s in ordered[1:]:
        assert any(v in bound for v in s.pattern.variables())
        bound.update(s.pattern.variables())


def test_star_subgroup_detection():
    st = Streamertail(chain_stats())
    scans = [_scan("?e", 1, "?a"), _scan("?e", 2, "?b"), _scan("?e", 3, "?c")]
    star = st._star_subgroup(scans, set())
    assert star is not None
    assert all(s.pattern.s.name == "e" for s in star[:3])
    # ascending by estimated size: predicate 3 (100 rows) first
    assert star[0].pattern.p.id == 3


def test_no_shared_vars_gives_nested_loop():
    st = Streamertail(chain_stats())
    plan, _, _ = st._plan_scan_group(
        [_scan("?a", 1, "?b"), _scan("?x", 3, "?y")], set())
    assert isinstance(plan, PNestedLoopJoin)


def test_bound_scan_estimates_discount():
    est = CostEstimator(chain_stats())
    pat = TriplePattern(Variable("s"), Constant(1), Variable("o"))
    free = est.estimate_scan(pat, set())
    bound = est.estimate_scan(pat, {"s"})
    assert free == 50_000
    assert bound == pytest.approx(1.0)  # 50k rows / 50k distinct subjects


def test_scan_choice_by_boundness():
    st = Streamertail(chain_stats())
    pat_bound = TriplePattern(Constant(7), Constant(1), Variable("o"))
    op = st._choose_scan(pat_bound, set(), None)
    assert isinstance(op, PIndexScan)
    pat_free = TriplePattern(Variable("s"), Variable("p"), Variable("o"))
    op = st._choose_scan(pat_free, set(), None)
    assert isinstance(op, PTableScan)


def test_join_shapes_agree_on_results():
    """Result-set equality across forced join shapes (ref
    join_ordering_shapes_test.rs)."""
    import os
    from kolibrie_amd import SparqlDatabase
    EX = "http://e/"
    db = SparqlDatabase()
    for i in range(30):
        db.add_triple(f"<{EX}e{i}>", f"<{EX}worksFor>", f"<{EX}d{i % 5}>")
        db.add_triple(f"<{EX}e{i}>", f"<{EX}sal>", f'"{1000 + i}"')
    for d in range(5):
        db.add_triple(f"<{EX}d{d}>", f"<{EX}city>", f'"c{d % 2}"')
    q = f"""SELECT ?e ?c WHERE {{
        ?e <{EX}worksFor> ?d . ?e <{EX}sal> ?s . ?d <{EX}city> ?c }}"""
    results = {}
    for mode in ("auto", "hash", "bind"):
        os.environ["KOLIBRIE_JOIN_MODE"] = mode
        db._plan_cache = {}
        results[mode] = sorted(map(tuple, db.query(q)))
    os.environ.pop("KOLIBRIE_JOIN_MODE", None)
    assert results["auto"] == results["hash"] == results["bind"]
    assert len(results["auto"]) == 30


def test_stats_index_path_matches_torch_oracle():
    """Single-graph stats come from sorted-order transition counts; they
    must equal the unique/segment oracle (and the multi-graph path)."""
    import torch
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.plan.stats import DatabaseStats

    torch.manual_seed(11)
    db = SparqlDatabase(device="cpu")
    n = 5000
    s = torch.randint(0, 200, (n,), dtype=torch.int32)
    p = torch.randint(0, 9, (n,), dtype=torch.int32)
    o = torch.randint(0, 400, (n,), dtype=torch.int32)
    db.store.insert_bulk(0, s, p, o)

    st = DatabaseStats.gather(db)
    # oracle over the deduplicated committed columns
    cs, cp, co = db.store.graph_index(0).columns()
    oracle = DatabaseStats()
    oracle._gather_torch(cs, cp, co)
    assert st.pred_count == oracle.pred_count
    assert st.pred_distinct_subj == oracle.pred_distinct_subj
    assert st.pred_distinct_obj == oracle.pred_distinct_obj
    assert st.distinct_subjects == oracle.distinct_subjects
    assert st.distinct_objects == oracle.distinct_objects
