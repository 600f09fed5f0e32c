"""Provenance semirings, tagged fixpoint, TagStore, SDD, diff-SDD, hybrid
(mirrors datalog/tests/reasoning_tests.rs provenance tests and
kolibrie/tests/hybrid_test.rs)."""
import math

import pytest

from kolibrie_amd.reasoning.provenance import (
    AddMultProbability, BooleanProvenance, DnfWmcProvenance,
    ExpirationProvenance, MinMaxProbability, TopKProofs, semiring_by_name,
)
from kolibrie_amd.reasoning.provenance_fixpoint import infer_with_provenance
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def _tp(s, p, o):
    def t(x):
        return Variable(x[1:]) if isinstance(x, str) and x.startswith("?") \
            else Constant(x)
    return TriplePattern(t(s), t(p), t(o))


# ----------------------------------------------------------------- semirings
def test_minmax_semiring():
    s = MinMaxProbability()
    assert s.plus(0.3, 0.7) == 0.7
    assert s.times(0.3, 0.7) == 0.3
    assert s.negate(0.3) == pytest.approx(0.7)
    assert s.zero() == 0.0 and s.one() == 1.0


def test_addmult_semiring():
    s = AddMultProbability()
    # noisy-or disjunction (ref provenance.rs:119): a+b-ab
    assert abs(s.plus(0.6, 0.7) - 0.88) < 1e-9
    assert s.times(0.5, 0.5) == 0.25


def test_topk_proofs():
    s = TopKProofs(k=2)
    a = s.tag_from_probability(0.9, seed_id=1)
    b = s.tag_from_probability(0.5, seed_id=2)
    c = s.tag_from_probability(0.4, seed_id=3)
    merged = s.plus(s.plus(a, b), c)
    assert len(merged) == 2  # truncated to k best
    assert merged[0] == frozenset([1])
    # times = cartesian union
    prod = s.times(a, b)
    assert prod == (frozenset([1, 2]),)
    assert s.proof_probability(prod[0]) == pytest.approx(0.45)
    # inclusion-exclusion: P(1 or 2) = .9 + .5 - .45
    assert s.recover(s.plus(a, b)) == pytest.approx(0.95)


def test_topk_subsumption():
    s = TopKProofs(k=4)
    a = s.tag_from_probability(0.9, seed_id=1)
    ab = s.times(a, s.tag_from_probability(0.8, seed_id=2))
    merged = s.plus(a, ab)
    assert merged == (frozenset([1]),)  # superset proof subsumed


def test_dnf_wmc():
    s = DnfWmcProvenance()
    a = s.tag_from_probability(0.5, seed_id=1)
    b = s.tag_from_probability(0.5, seed_id=2)
    either = s.plus(a, b)
    assert s.recover(either) == pytest.approx(0.75)
    both = s.times(a, b)
    assert s.recover(both) == pytest.approx(0.25)
    neg = s.negate(either)
    assert s.recover(neg) == pytest.approx(0.25)
    # contradiction pruning: a AND NOT a = 0
    contra = s.times(a, s.negate(a))
    assert s.recover(contra) == 0.0


def test_expiration_semiring():
    s = ExpirationProvenance()
    assert s.times(10.0, 15.0) == 10.0
    assert s.plus(10.0, 15.0) == 15.0


def test_semiring_by_name():
    assert semiring_by_name("minmax").name == "minmax"
    assert semiring_by_name("independent").name == "addmult"
    with pytest.raises(ValueError):
        semiring_by_name("nope")


# ------------------------------------------------------------ tagged fixpoint
def test_provenance_fixpoint_minmax():
    P, Q = 100, 101
    a, b, c = 1, 2, 3
    rule = Rule(
        premise=[_tp("?x", P, "?y"), _tp("?y", P, "?z")],
        conclusion=[_tp("?x", Q, "?z")],
    )
    sr = MinMaxProbability()
    seeds = {(a, P, b): 0.9, (b, P, c): 0.6}
    known = infer_with_provenance([rule], seeds, sr)
    assert known[(a, Q, c)] == pytest.approx(0.6)  # min along the chain


def test_provenance_fixpoint_tag_improvement_reenters_delta():
    # two derivations of the same fact with different strengths: the
    # stronger one must propagate (delta_improved semantics)
    P, Q, R = 100, 101, 102
    a, b, c, d = 1, 2, 3, 4
    rules = [
        Rule(premise=[_tp("?x", P, "?y")], conclusion=[_tp("?x", Q, "?y")]),
        Rule(premise=[_tp("?x", Q, "?y"), _tp("?y", Q, "?z")],
             conclusion=[_tp("?x", R, "?z")]),
    ]
    sr = MinMaxProbability()
    seeds = {
        (a, P, b): 0.3,
        (b, P, c): 0.9,
        (a, Q, b): 0.8,   # stronger direct assertion of the derived fact
    }
    known = infer_with_provenance(rules, seeds, sr)
    # (a R c) = min(max(0.3, 0.8), 0.9) = 0.8 — requires the improved
    # (a Q b)=0.8 tag to re-enter the delta
    assert known[(a, R, c)] == pytest.approx(0.8)


def test_provenance_fixpoint_stratified_naf():
    P, B, Q = 100, 101, 102
    a, b = 1, 2
    rule = Rule(
        premise=[_tp("?x", P, "?y")],
        negative_premise=[_tp("?x", B, "?y")],
        conclusion=[_tp("?x", Q, "?y")],
    )
    sr = MinMaxProbability()
    known = infer_with_provenance(
        [rule], {(a, P, b): 0.9, (a, B, b): 0.3}, sr)
    # negated premise present with tag 0.3 -> factor 0.7: min(0.9, 0.7)
    assert known[(a, Q, b)] == pytest.approx(0.7)


# ------------------------------------------------------------------ TagStore
def test_tag_store_roundtrip():
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.reasoning.tag_store import TagStore
    sr = MinMaxProbability()
    ts = TagStore(sr)
    db = SparqlDatabase()
    s = db.dictionary.encode("s")
    p = db.dictionary.encode("p")
    o = db.dictionary.encode("o")
    assert ts.update_disjunction((s, p, o), 0.4)
    assert ts.update_disjunction((s, p, o), 0.7)
    assert not ts.update_disjunction((s, p, o), 0.5)  # max unchanged
    assert ts.probability((s, p, o)) == pytest.approx(0.7)
    ts.encode_into_db(db)
    back = TagStore.decode_from_db(db, sr)
    assert back.probability((s, p, o)) == pytest.approx(0.7)


# ----------------------------------------------------------------------- SDD
def test_sdd_basic_wmc():
    from kolibrie_amd.reasoning.sdd import SddManager
    m = SddManager()
    m.declare_var(1, 0.5)
    m.declare_var(2, 0.5)
    a = m.literal(1)
    b = m.literal(2)
    assert m.wmc(m.disjoin(a, b)) == pytest.approx(0.75)
    assert m.wmc(m.conjoin(a, b)) == pytest.approx(0.25)
    assert m.wmc(m.negate(a)) == pytest.approx(0.5)
    assert m.wmc(m.conjoin(a, m.negate(a))) == 0.0


def test_sdd_exactly_one():
    from kolibrie_amd.reasoning.sdd import SddManager
    m = SddManager()
    for v, p in ((1, 0.2), (2, 0.3), (3, 0.5)):
        m.declare_var(v, p)
    node = m.exactly_one([1, 2, 3])
    expect = 0.2 * 0.7 * 0.5 + 0.8 * 0.3 * 0.5 + 0.8 * 0.7 * 0.5
    assert m.wmc(node) == pytest.approx(expect)
    models = list(m.models(node))
    assert len(models) == 3


def test_sdd_budget():
    from kolibrie_amd.reasoning.sdd import SddManager, SddOperationBudget
    m = SddManager()
    big1 = m.true_node()
    big2 = m.true_node()
    for v in range(1, 25):
        m.declare_var(v, 0.5)
    import functools
    # xor chains blow up node count under a fixed order
    for v in range(1, 13):
        big1 = m.apply("xor", big1, m.literal(v))
    budget = SddOperationBudget(max_nodes=m.node_count() + 5)
    res = m.try_apply("xor", big1, m.literal(20), budget)
    # tiny cap: the op must refuse rather than blow the budget
    assert res is None or m.node_count() <= budget.max_nodes + 2


def test_diff_sdd_gradient():
    from kolibrie_amd.reasoning.diff_sdd import wmc_gradient
    from kolibrie_amd.reasoning.sdd import SddManager
    m = SddManager()
    m.declare_var(1, 0.3)
    m.declare_var(2, 0.6)
    f = m.disjoin(m.literal(1), m.literal(2))   # P = p1 + p2 - p1 p2
    g = wmc_gradient(m, f)
    assert g[1] == pytest.approx(1 - 0.6)   # dP/dp1 = 1 - p2
    assert g[2] == pytest.approx(1 - 0.3)
    # finite-difference check
    eps = 1e-6
    m.pos_weight[1] = 0.3 + eps
    m.neg_weight[1] = 0.7 - eps
    p_hi = m.wmc(f)
    m.pos_weight[1] = 0.3
    m.neg_weight[1] = 0.7
    assert (p_hi - m.wmc(f)) / eps == pytest.approx(g[1], rel=1e-3)


# -------------------------------------------------------------------- hybrid
def test_hybrid_config_validation():
    from kolibrie_amd.reasoning.hybrid import HybridConfig
    HybridConfig().validate()
    with pytest.raises(ValueError):
        HybridConfig(threshold=1.5).validate()
    with pytest.raises(ValueError):
        HybridConfig(k_initial=0).validate()


def test_hybrid_lineage_store_hashing():
    from kolibrie_amd.reasoning.hybrid import LineageStore
    st = LineageStore()
    l1, l2 = st.leaf(1), st.leaf(2)
    a1 = st.and_node([l1, l2])
    a2 = st.and_node([l2, l1])
    assert a1 == a2  # structural hashing, order-insensitive
    o1 = st.or_node([a1, l1])
    assert st.or_node([l1, a1]) == o1


def test_hybrid_evaluate_decides():
    from kolibrie_amd.reasoning.hybrid import (
        HybridConfig, LineageStore, evaluate_hybrid,
    )
    st = LineageStore()
    node = st.or_node([st.leaf(1), st.leaf(2)])
    weights = {1: 0.9, 2: 0.8}
    res = evaluate_hybrid(st, node, weights, HybridConfig(threshold=0.5))
    assert res.above_threshold is True
    assert res.probability == pytest.approx(0.98, abs=0.01)


def test_hybrid_escalates_on_negation():
    from kolibrie_amd.reasoning.hybrid import (
        HybridConfig, LineageStore, evaluate_hybrid,
    )
    st = LineageStore()
    node = st.and_node([st.leaf(1), st.not_node(st.leaf(2))])
    weights = {1: 0.9, 2: 0.5}
    res = evaluate_hybrid(st, node, weights, HybridConfig(threshold=0.5))
    assert res.metrics.escalated
    assert res.probability == pytest.approx(0.45)
    assert res.status == "DecidedExact"


def test_hybrid_rejects_nonmonotone_rules():
    from kolibrie_amd.reasoning.hybrid import validate_monotone
    r = Rule(premise=[_tp("?x", 1, "?y")],
             negative_premise=[_tp("?x", 2, "?y")],
             conclusion=[_tp("?x", 3, "?y")])
    with pytest.raises(ValueError):
        validate_monotone([r])


def test_hybrid_materialize_and_evaluate():
    from kolibrie_amd.reasoning.hybrid import (
        HybridConfig, evaluate_hybrid, materialize_lineage,
    )
    P, Q = 100, 101
    a, b, c = 1, 2, 3
    rule = Rule(premise=[_tp("?x", P, "?y"), _tp("?y", P, "?z")],
                conclusion=[_tp("?x", Q, "?z")])
    store, nodes, weights = materialize_lineage(
        [rule], {(a, P, b): 0.9, (b, P, c): 0.8})
    res = evaluate_hybrid(store, nodes[(a, Q, c)], weights,
                          HybridConfig(threshold=0.5))
    assert res.probability == pytest.approx(0.72, abs=0.02)
    assert res.above_threshold is True


def test_hybrid_fake_clock_budget():
    from kolibrie_amd.reasoning.hybrid import (
        FakeClock, HybridConfig, LineageStore, evaluate_hybrid,
    )
    st = LineageStore()
    node = st.or_node([st.leaf(i) for i in range(1, 6)])
    weights = {i: 0.1 for i in range(1, 6)}
    clock = FakeClock()
    clock.advance(1000.0)  # already past every deadline at entry
    cfg = HybridConfig(threshold=0.5, topk_budget_ms=0.0)
    res = evaluate_hybrid(st, node, weights, cfg, clock=clock)
    assert res.status in ("Decided", "DecidedExact", "Inconclusive")


def test_sdd_wmc_vs_bruteforce_random():
    """SDD WMC vs exhaustive model enumeration on 60 random weighted DNFs
    (<= 7 vars): exact agreement."""
    import itertools
    import random
    from kolibrie_amd.reasoning.sdd import SddManager
    rng = random.Random(99)
    for _ in range(60):
        nvars = rng.randint(2, 7)
        weights = {v: round(rng.uniform(0.05, 0.95), 3)
                   for v in range(1, nvars + 1)}
        m = SddManager()
        for v, w in weights.items():
            m.declare_var(v, pos_weight=w, neg_weight=1.0 - w)
        node = m.false_node()
        clauses = []
        for _ in range(rng.randint(1, 4)):
            lits = [(rng.randint(1, nvars), rng.random() < 0.5)
                    for _ in range(rng.randint(1, 3))]
            clauses.append(lits)
            c = m.true_node()
            for v, pos in lits:
                c = m.conjoin(c, m.literal(v, pos))
            node = m.disjoin(node, c)
        got = m.wmc(node)
        want = 0.0
        for assign in itertools.product([False, True], repeat=nvars):
            a = {v: assign[v - 1] for v in range(1, nvars + 1)}
            if any(all(a[v] == pos for v, pos in cl) for cl in clauses):
                p = 1.0
                for v in range(1, nvars + 1):
                    p *= weights[v] if a[v] else (1.0 - weights[v])
                want += p
        assert abs(got - want) < 1e-9, (clauses, got, want)


def test_wmc_gradient_vs_finite_differences():
    """diff-SDD gradients agree with central finite differences over
    random weighted circuits."""
    import random
    from kolibrie_amd.reasoning.diff_sdd import wmc_gradient
    from kolibrie_amd.reasoning.sdd import SddManager
    rng = random.Random(5)
    for trial in range(30):
        nvars = rng.randint(2, 6)
        weights = {v: round(rng.uniform(0.1, 0.9), 3)
                   for v in range(1, nvars + 1)}

        def build(ws):
            m = SddManager()
            for v, w in ws.items():
                m.declare_var(v, pos_weight=w, neg_weight=1.0 - w)
            node = m.false_node()
            r2 = random.Random(trial)
            for _ in range(r2.randint(1, 3)):
                c = m.true_node()
                for _ in range(r2.randint(1, 3)):
                    c = m.conjoin(c, m.literal(r2.randint(1, nvars),
                                               r2.random() < 0.5))
                node = m.disjoin(node, c)
            return m, node

        m, node = build(weights)
        grads = wmc_gradient(m, node)
        eps = 1e-6
        for v in range(1, nvars + 1):
            wp = dict(weights); wp[v] += eps
            wm = dict(weights); wm[v] -= eps
            mp, np_ = build(wp)
            mm, nm = build(wm)
            fd = (mp.wmc(np_) - mm.wmc(nm)) / (2 * eps)
            assert abs(grads.get(v, 0.0) - fd) < 1e-4, (trial, v)


def test_hybrid_probability_vs_bruteforce_random():
    """Hybrid escalation's exact probabilities vs brute-force enumeration
    over random seed assignments (25 random 2-rule programs)."""
    import itertools
    import random
    from kolibrie_amd.reasoning.hybrid import (HybridConfig,
                                               evaluate_hybrid,
                                               materialize_lineage)
    from kolibrie_amd.reasoning.rule import Rule
    from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable

    def _tp2(s, p, o):
        t = lambda x: (Variable(x[1:]) if isinstance(x, str)
                       and x.startswith("?") else Constant(x))
        return TriplePattern(t(s), t(p), t(o))

    P2, Q2, R2 = 100, 101, 102
    rng = random.Random(8)
    for trial in range(25):
        rules = [
            Rule(premise=[_tp2("?x", P2, "?y")],
                 conclusion=[_tp2("?x", Q2, "?y")]),
            Rule(premise=[_tp2("?x", Q2, "?y"), _tp2("?y", Q2, "?z")],
                 conclusion=[_tp2("?x", R2, "?z")]),
        ]
        seeds = {(rng.randint(1, 5), P2, rng.randint(1, 5)):
                 round(rng.uniform(0.1, 0.9), 3)
                 for _ in range(rng.randint(2, 6))}
        store, node_by_triple, weights = materialize_lineage(
            rules, dict(seeds), set())
        cfg = HybridConfig(threshold=0.5)
        seed_list = sorted(weights)
        for t, tag in node_by_triple.items():
            if t in seeds or tag == -1:
                continue
            res = evaluate_hybrid(store, tag, weights, cfg)
            if res.probability is None:
                continue
            want = 0.0
            for assign in itertools.product([False, True],
                                            repeat=len(seed_list)):
                amap = dict(zip(seed_list, assign))
                memo = {}

                def ev(nid):
                    if nid in memo:
                        return memo[nid]
                    kind, ch = store.nodes[nid]
                    if kind == "true":
                        r = True
                    elif kind == "leaf":
                        r = amap[ch[0]]
                    elif kind == "and":
                        r = all(ev(c) for c in ch)
                    elif kind == "or":
                        r = any(ev(c) for c in ch)
                    else:
                        r = not ev(ch[0])
                    memo[nid] = r
                    return r

                if ev(tag):
                    p = 1.0
                    for sid, val in amap.items():
                        w = weights[sid]
                        p *= w if val else (1.0 - w)
                    want += p
            assert abs(res.probability - want) < 1e-6, (trial, t)
