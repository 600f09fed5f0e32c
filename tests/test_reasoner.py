"""Datalog reasoning (mirrors datalog/tests/reasoning_tests.rs: naive vs
semi-naive equality, NAF, transitive closure, constraints/repairs)."""
import pytest

from kolibrie_amd import Reasoner
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def _c(r, s):
    x = r.dictionary.encode(s) & 0xFFFFFFFF
    return Constant(x - 0x1_0000_0000 if x >= 0x8000_0000 else x)


def _tp(r, s, p, o):
    def t(x):
        if isinstance(x, str) and x.startswith("?"):
            return Variable(x[1:])
        return _c(r, x)
    return TriplePattern(t(s), t(p), t(o))


def test_transitive_closure_semi_naive():
    r = Reasoner()
    for i in range(10):
        r.add_abox_triple(f"n{i}", "edge", f"n{i+1}")
    r.add_rule(Rule(
        premise=[_tp(r, "?x", "edge", "?y"), _tp(r, "?y", "reach", "?z")],
        conclusion=[_tp(r, "?x", "reach", "?z")],
    ))
    r.add_rule(Rule(
        premise=[_tp(r, "?x", "edge", "?y")],
        conclusion=[_tp(r, "?x", "reach", "?y")],
    ))
    n = r.infer_new_facts_semi_naive()
    # reach = all pairs i<j over 11 nodes: 55; 10 are direct edges
    assert n == 55
    assert r.contains_fact("n0", "reach", "n10")


def test_naive_equals_semi_naive():
    def build():
        r = Reasoner()
        r.add_abox_triple("a", "p", "b")
        r.add_abox_triple("b", "p", "c")
        r.add_abox_triple("c", "p", "d")
        r.add_rule(Rule(
            premise=[_tp(r, "?x", "p", "?y"), _tp(r, "?y", "p", "?z")],
            conclusion=[_tp(r, "?x", "q", "?z")],
        ))
        return r
    r1, r2 = build(), build()
    r1.infer_new_facts()
    r2.infer_new_facts_semi_naive()
    assert r1.all_fact_tuples() == r2.all_fact_tuples()


def test_rule_text_parsing():
    r = Reasoner()
    r.add_abox_triple("http://e/alice", "http://e/parent", "http://e/bob")
    r.add_abox_triple("http://e/bob", "http://e/parent", "http://e/carol")
    r.add_rule_text("""
        RULE :Grandparent :- CONSTRUCT { ?x <http://e/grandparent> ?z }
        WHERE { ?x <http://e/parent> ?y . ?y <http://e/parent> ?z } .
    """)
    r.infer_new_facts_semi_naive()
    assert r.contains_fact("http://e/alice", "http://e/grandparent", "http://e/carol")


def test_naf_negation():
    r = Reasoner()
    r.add_abox_triple("a", "item", "x")
    r.add_abox_triple("b", "item", "y")
    r.add_abox_triple("a", "blocked", "x")
    r.add_rule(Rule(
        premise=[_tp(r, "?s", "item", "?v")],
        negative_premise=[_tp(r, "?s", "blocked", "?v")],
        conclusion=[_tp(r, "?s", "allowed", "?v")],
    ))
    r.infer_new_facts_semi_naive()
    assert not r.contains_fact("a", "allowed", "x")
    assert r.contains_fact("b", "allowed", "y")


def test_rule_filters():
    r = Reasoner()
    r.add_abox_triple("m1", "temp", "95")
    r.add_abox_triple("m2", "temp", "50")
    r.add_rule_text("""
        RULE :Hot :- CONSTRUCT { ?m <alert> "hot" }
        WHERE { ?m <temp> ?t . FILTER(?t > 90) } .
    """)
    r.infer_new_facts_semi_naive()
    assert r.contains_fact("m1", "alert", "hot")
    assert not r.contains_fact("m2", "alert", "hot")


def test_query_abox():
    r = Reasoner()
    r.add_abox_triple("a", "p", "b")
    r.add_abox_triple("a", "q", "c")
    out = r.query_abox("a", None, None)
    assert out == [("a", "p", "b"), ("a", "q", "c")]
    assert r.query_abox(None, "q", None) == [("a", "q", "c")]


def test_constraints_and_repairs():
    r = Reasoner()
    r.add_abox_triple("x", "status", "on")
    r.add_abox_triple("x", "status", "off")
    # constraint: nothing may be both on and off
    r.add_constraint(Rule(
        premise=[_tp(r, "?s", "status", "on"), _tp(r, "?s", "status", "off")],
        conclusion=[],
    ))
    assert r.violates_constraints()
    repairs = r.compute_repairs()
    assert repairs
    assert all(len(rep) == 1 for rep in repairs)


def test_deep_taxonomy_scaling():
    """Deep-taxonomy shape: subclass chain depth 100."""
    r = Reasoner()
    depth = 100
    for i in range(depth):
        r.add_abox_triple(f"C{i}", "subClassOf", f"C{i+1}")
    r.add_abox_triple("i0", "type", "C0")
    r.add_rule(Rule(
        premise=[_tp(r, "?x", "type", "?c"), _tp(r, "?c", "subClassOf", "?d")],
        conclusion=[_tp(r, "?x", "type", "?d")],
    ))
    n = r.infer_new_facts_semi_naive()
    assert n == depth
    assert r.contains_fact("i0", "type", f"C{depth}")


def test_provenance_entry_on_reasoner():
    r = Reasoner()
    r.add_abox_triple("a", "p", "b")
    r.add_abox_triple("b", "p", "c")
    a = r.dictionary.encode("a")
    p = r.dictionary.encode("p")
    b = r.dictionary.encode("b")
    c = r.dictionary.encode("c")
    q = r.dictionary.encode("q")
    r.probability_seeds[(a, p, b)] = 0.9
    r.probability_seeds[(b, p, c)] = 0.6
    r.add_rule(Rule(
        premise=[_tp(r, "?x", "p", "?y"), _tp(r, "?y", "p", "?z")],
        conclusion=[_tp(r, "?x", "q", "?z")],
    ))
    tags = r.infer_new_facts_with_provenance("minmax")
    assert abs(tags[(a, q, c)] - 0.6) < 1e-6


def test_sdd_seeded_materialisation():
    r = Reasoner()
    r.add_abox_triple("a", "p", "b")
    a = r.dictionary.encode("a")
    p = r.dictionary.encode("p")
    b = r.dictionary.encode("b")
    q = r.dictionary.encode("q")
    r.probability_seeds[(a, p, b)] = 0.7
    r.add_rule(Rule(premise=[_tp(r, "?x", "p", "?y")],
                    conclusion=[_tp(r, "?x", "q", "?y")]))
    tags, prov = r.infer_with_sdd_seeds()
    assert abs(prov.recover(tags[(a, q, b)]) - 0.7) < 1e-6


def test_query_with_repairs_iar():
    r = Reasoner()
    r.add_abox_triple("x", "status", "on")
    r.add_abox_triple("x", "status", "off")
    r.add_abox_triple("y", "status", "on")
    r.add_constraint(Rule(
        premise=[_tp(r, "?s", "status", "on"), _tp(r, "?s", "status", "off")],
        conclusion=[],
    ))
    # y's status survives every repair; x's conflicting facts do not
    answers = r.query_with_repairs(None, "status", None)
    assert ("y", "status", "on") in answers
    assert ("x", "status", "on") not in answers
    assert ("x", "status", "off") not in answers


def test_window_runner():
    from kolibrie_amd.rsp.s2r import CSPARQLWindow, Report, ReportStrategy, Tick
    from kolibrie_amd.rsp.window_runner import WindowRunner
    rep = Report()
    rep.add(ReportStrategy.ON_WINDOW_CLOSE)
    runner = WindowRunner(CSPARQLWindow(5, 5, rep, Tick.TIME_DRIVEN, "w"))
    for ts in range(0, 11):
        runner.push(("e", ts), ts)
    fired = runner.drain()
    assert len(fired) == 2


# ---- forward-chaining shape parity (ref datalog/tests/reasoning_tests.rs:
# fc_variable_predicate_premise, fc_repeated_variable_premise,
# fc_constant_subject/object_premise, fc_multi_conclusion,
# fc_diamond_ancestor, fc_shared_predicate_variable, fc_idempotent) ----

def _rule(r, concl, body):
    r.add_rule_text(f"RULE :t :- CONSTRUCT {{ {concl} }} WHERE {{ {body} }}")


def test_fc_variable_predicate_premise():
    r = Reasoner()
    r.add_abox_triple("a", "likes", "b")
    r.add_abox_triple("b", "hates", "c")
    _rule(r, "?x <related> ?y", "?x ?p ?y")
    r.infer_new_facts_semi_naive()
    assert sorted(r.query_abox(None, "related", None)) == [
        ("a", "related", "b"), ("b", "related", "c")]


def test_fc_repeated_variable_premise():
    r = Reasoner()
    r.add_abox_triple("a", "p", "a")
    r.add_abox_triple("a", "p", "b")
    _rule(r, "?x <selfloop> ?x", "?x <p> ?x")
    r.infer_new_facts_semi_naive()
    assert r.query_abox(None, "selfloop", None) == [("a", "selfloop", "a")]


def test_fc_shared_predicate_variable_join():
    r = Reasoner()
    r.add_abox_triple("a", "q", "b")
    r.add_abox_triple("b", "q", "c")
    r.add_abox_triple("b", "r", "d")  # different predicate: must not join
    _rule(r, "?x <two> ?z", "?x ?p ?y . ?y ?p ?z")
    r.infer_new_facts_semi_naive()
    assert r.query_abox(None, "two", None) == [("a", "two", "c")]


def test_fc_constant_subject_premise_filters():
    r = Reasoner()
    r.add_abox_triple("admin", "grants", "u1")
    r.add_abox_triple("other", "grants", "u2")
    _rule(r, "?u <trusted> <yes>", "<admin> <grants> ?u")
    r.infer_new_facts_semi_naive()
    assert r.query_abox(None, "trusted", None) == [("u1", "trusted", "yes")]


def test_fc_multi_conclusion():
    r = Reasoner()
    r.add_abox_triple("x", "parent", "y")
    _rule(r, "?b <child> ?a . ?a <ancestor> ?b", "?a <parent> ?b")
    r.infer_new_facts_semi_naive()
    assert r.query_abox(None, "child", None) == [("y", "child", "x")]
    assert r.query_abox(None, "ancestor", None) == [("x", "ancestor", "y")]


def test_fc_diamond_ancestor_dedup():
    r = Reasoner()
    for s, p, o in [("top", "edge", "l"), ("top", "edge", "r"),
                    ("l", "edge", "bot"), ("r", "edge", "bot")]:
        r.add_abox_triple(s, p, o)
    _rule(r, "?x <reach> ?z", "?x <edge> ?z")
    _rule(r, "?x <reach> ?z", "?x <reach> ?y . ?y <edge> ?z")
    r.infer_new_facts_semi_naive()
    # both diamond paths derive top->bot exactly once
    assert sorted(r.query_abox("top", "reach", None)) == [
        ("top", "reach", "bot"), ("top", "reach", "l"), ("top", "reach", "r")]


def test_fc_idempotent_reinfer():
    r = Reasoner()
    r.add_abox_triple("a", "edge", "b")
    _rule(r, "?x <reach> ?y", "?x <edge> ?y")
    n1 = r.infer_new_facts_semi_naive()
    n2 = r.infer_new_facts_semi_naive()
    assert n1 >= 1 and n2 == 0


def test_infer_with_repairs_restores_consistency():
    """ref reasoning.rs infer_new_facts_semi_naive_with_repairs: after the
    fixpoint, a minimal repair removes a violating base fact."""
    r = Reasoner()
    r.add_abox_triple("x", "status", "active")
    r.add_abox_triple("x", "status", "banned")
    r.add_abox_triple("y", "status", "active")
    from kolibrie_amd.reasoning.rule import Rule
    from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable
    status = r.dictionary.encode("status")
    active = r.dictionary.encode("active")
    banned = r.dictionary.encode("banned")
    constraint = Rule(
        premise=[TriplePattern(Variable("u"), Constant(status), Constant(active)),
                 TriplePattern(Variable("u"), Constant(status), Constant(banned))],
        conclusion=[],
    )
    r.add_constraint(constraint)
    assert r.violates_constraints()
    r.infer_new_facts_semi_naive_with_repairs()
    assert not r.violates_constraints()
    # y's fact untouched
    assert ("y", "status", "active") in r.query_abox("y", None, None)


def test_streamertail_execute_plan_api():
    """ref README Streamertail::execute_plan returning var->value maps."""
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.parsing.sparql import parse_combined_query
    from kolibrie_amd.plan.lower import build_logical_plan
    from kolibrie_amd.plan.optimizer import Streamertail
    db = SparqlDatabase()
    db.add_triple("<http://e/a>", "<http://e/p>", "<http://e/b>")
    cq = parse_combined_query("SELECT ?s ?o WHERE { ?s <http://e/p> ?o }")
    logical = build_logical_plan(cq.select.where, db, {})
    st = Streamertail.with_cached_stats(db.get_or_build_stats())
    plan = st.find_best_plan(logical)
    rows = st.execute_plan(plan, db)
    assert rows == [{"s": "http://e/a", "o": "http://e/b"}]


def test_add_tbox_triple_participates_in_inference():
    r = Reasoner()
    r.add_tbox_triple("Dog", "subClassOf", "Animal")
    r.add_abox_triple("rex", "type", "Dog")
    r.add_rule_text(
        "RULE :sc :- CONSTRUCT { ?x <type> ?super } "
        "WHERE { ?x <type> ?c . ?c <subClassOf> ?super }")
    r.infer_new_facts_semi_naive()
    assert ("rex", "type", "Animal") in r.query_abox("rex", None, None)
