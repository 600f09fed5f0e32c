"""Backward chaining, proof export and hierarchical reasoning
(mirrors datalog/tests backward-chaining tests and the
hierarchy_reasoning examples)."""
import pytest

from kolibrie_amd import Reasoner
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def _tp(r, s, p, o):
    def t(x):
        if isinstance(x, str) and x.startswith("?"):
            return Variable(x[1:])
        x = r.dictionary.encode(x) & 0xFFFFFFFF
        return Constant(x - 0x1_0000_0000 if x >= 0x8000_0000 else x)
    return TriplePattern(t(s), t(p), t(o))


def test_backward_chaining_facts_only():
    r = Reasoner()
    r.add_abox_triple("alice", "knows", "bob")
    r.add_abox_triple("alice", "knows", "carol")
    res = r.backward_chaining(("alice", "knows", "?who"))
    who = sorted(r.dictionary.decode(b["who"]) for b in res)
    assert who == ["bob", "carol"]


def test_backward_chaining_through_rule():
    r = Reasoner()
    r.add_abox_triple("alice", "parent", "bob")
    r.add_abox_triple("bob", "parent", "carol")
    r.add_rule(Rule(
        premise=[_tp(r, "?x", "parent", "?y"), _tp(r, "?y", "parent", "?z")],
        conclusion=[_tp(r, "?x", "grandparent", "?z")],
    ))
    res = r.backward_chaining(("?g", "grandparent", "carol"))
    assert [r.dictionary.decode(b["g"]) for b in res] == ["alice"]
    # ground goal provable
    assert r.backward_chaining(("alice", "grandparent", "carol")) != []
    assert r.backward_chaining(("bob", "grandparent", "carol")) == []


def test_backward_chaining_recursive_rule_terminates():
    r = Reasoner()
    for i in range(5):
        r.add_abox_triple(f"n{i}", "edge", f"n{i+1}")
    r.add_rule(Rule(premise=[_tp(r, "?x", "edge", "?y")],
                    conclusion=[_tp(r, "?x", "reach", "?y")]))
    r.add_rule(Rule(
        premise=[_tp(r, "?x", "edge", "?y"), _tp(r, "?y", "reach", "?z")],
        conclusion=[_tp(r, "?x", "reach", "?z")],
    ))
    res = r.backward_chaining(("n0", "reach", "?t"), max_depth=10)
    targets = sorted(r.dictionary.decode(b["t"]) for b in res)
    assert targets == [f"n{i}" for i in range(1, 6)]


def test_to_dot_export():
    from kolibrie_amd.reasoning.to_dot import facts_to_dot, proof_graph_to_dot
    r = Reasoner()
    r.add_abox_triple("a", "p", "b")
    dot = facts_to_dot(r)
    assert dot.startswith("digraph") and '"a" -> "b"' in dot
    a = r.dictionary.encode("a")
    p = r.dictionary.encode("p")
    b = r.dictionary.encode("b")
    q = r.dictionary.encode("q")
    proof = proof_graph_to_dot(r, {(a, q, b): [(a, p, b)]})
    assert "->" in proof and "ellipse" in proof


def test_reasoning_hierarchy():
    from kolibrie_amd.reasoning.hierarchy import (
        HierarchicalRule, ReasoningHierarchy, ReasoningLevel,
    )
    h = ReasoningHierarchy()
    h.add_fact(ReasoningLevel.BASE, "x", "is", "mammal")
    base = h.levels[ReasoningLevel.BASE]
    h.add_rule(HierarchicalRule(
        rule=Rule(premise=[_tp(base, "?a", "is", "mammal")],
                  conclusion=[_tp(base, "?a", "is", "animal")]),
        level=ReasoningLevel.BASE))
    h.add_rule(HierarchicalRule(
        rule=Rule(premise=[_tp(base, "?a", "is", "animal")],
                  conclusion=[_tp(base, "?a", "needs", "oxygen")]),
        level=ReasoningLevel.DEDUCTIVE,
        depends_on=[ReasoningLevel.BASE]))
    n = h.hierarchical_inference()
    assert n >= 2
    ded = h.query_level(ReasoningLevel.DEDUCTIVE, "x", "needs", None)
    assert ded == [("x", "needs", "oxygen")]


# ---- bc_* shape parity (ref reasoning_tests.rs bc_direct_fact ..
# bc_no_spurious_negative) ----

def _bc_reasoner():
    r = Reasoner()
    for s, p, o in [("a", "parent", "b"), ("b", "parent", "c"),
                    ("c", "parent", "d"), ("x", "parent", "c")]:
        r.add_abox_triple(s, p, o)
    r.add_rule_text(
        "RULE :anc :- CONSTRUCT { ?x <ancestor> ?y } WHERE { ?x <parent> ?y }")
    r.add_rule_text(
        "RULE :anc2 :- CONSTRUCT { ?x <ancestor> ?z } "
        "WHERE { ?x <parent> ?y . ?y <ancestor> ?z }")
    return r


def test_bc_direct_fact():
    r = _bc_reasoner()
    assert r.backward_chaining(("a", "parent", "b")) != []


def test_bc_3hop_transitive():
    r = _bc_reasoner()
    assert r.backward_chaining(("a", "ancestor", "d")) != []


def test_bc_specific_target_and_no_result():
    r = _bc_reasoner()
    assert r.backward_chaining(("a", "ancestor", "c")) != []
    assert r.backward_chaining(("d", "ancestor", "a")) == []
    assert r.backward_chaining(("a", "unknownpred", "b")) == []


def test_bc_full_scan_enumerates_all():
    r = _bc_reasoner()
    res = r.backward_chaining(("?s", "ancestor", "?o"))
    dec = r.dictionary.decode
    pairs = {(dec(b["s"]), dec(b["o"])) for b in res}
    assert ("a", "d") in pairs and ("x", "d") in pairs
    assert ("d", "a") not in pairs  # bc_no_spurious_negative


def test_bc_sibling_join():
    r = Reasoner()
    r.add_abox_triple("p", "childOf", "f")
    r.add_abox_triple("q", "childOf", "f")
    r.add_rule_text(
        "RULE :sib :- CONSTRUCT { ?a <sibling> ?b } "
        "WHERE { ?a <childOf> ?f . ?b <childOf> ?f }")
    res = r.backward_chaining(("p", "sibling", "?x"))
    assert {r.dictionary.decode(b["x"]) for b in res} >= {"q"}
