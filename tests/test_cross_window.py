"""Cross-window SDS+ parity (ref datalog/tests/cross_window_tests.rs:
hotspot derivation, naive==incremental, expiration times, expiry after a
supporting fact leaves, long-lived map facts, expiry chain propagation)."""
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.reasoning.sds import (
    Sds, WindowedTriple, incremental_sds_plus, naive_sds_plus,
)
from kolibrie_amd.storage.database import SparqlDatabase
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def _setup():
    db = SparqlDatabase()
    enc = db.dictionary.encode
    ids = {k: enc(k) for k in
           ("temp", "loc", "hot", "hotspot", "room", "zone", "reach")}
    return db, ids


def _rule(prem, concl):
    return Rule(premise=prem, conclusion=[concl])


def test_hotspot_two_window_join():
    """sensor temp (short window) x sensor location (long window) -> hotspot."""
    db, I = _setup()
    s1 = db.dictionary.encode("s1")
    r1 = db.dictionary.encode("r1")
    rule = _rule(
        [TriplePattern(Variable("s"), Constant(I["temp"]), Constant(I["hot"])),
         TriplePattern(Variable("s"), Constant(I["loc"]), Variable("r"))],
        TriplePattern(Variable("r"), Constant(I["hotspot"]), Constant(I["hot"])))
    sds = Sds()
    sds.add(WindowedTriple("wTemp", (s1, I["temp"], I["hot"]), 2), 5)
    sds.add(WindowedTriple("wLoc", (s1, I["loc"], r1), 0), 100)
    alive = naive_sds_plus(sds, [rule], db, 4)
    assert (r1, I["hotspot"], I["hot"]) in alive


def test_naive_incremental_agree_over_time():
    db, I = _setup()
    s1, r1 = db.dictionary.encode("s1"), db.dictionary.encode("r1")
    rule = _rule(
        [TriplePattern(Variable("s"), Constant(I["temp"]), Constant(I["hot"])),
         TriplePattern(Variable("s"), Constant(I["loc"]), Variable("r"))],
        TriplePattern(Variable("r"), Constant(I["hotspot"]), Constant(I["hot"])))
    sds = Sds()
    sds.add(WindowedTriple("wTemp", (s1, I["temp"], I["hot"]), 2), 5)
    sds.add(WindowedTriple("wLoc", (s1, I["loc"], r1), 1), 50)
    for ts in range(0, 60, 3):
        assert naive_sds_plus(sds, [rule], db, ts) == \
            incremental_sds_plus(sds, [rule], db, ts), ts


def test_derived_expiry_is_min_of_support():
    """Derived hotspot dies when the SHORT-window temp fact expires (at
    2+5=7), even though the location fact lives to 100."""
    db, I = _setup()
    s1, r1 = db.dictionary.encode("s1"), db.dictionary.encode("r1")
    rule = _rule(
        [TriplePattern(Variable("s"), Constant(I["temp"]), Constant(I["hot"])),
         TriplePattern(Variable("s"), Constant(I["loc"]), Variable("r"))],
        TriplePattern(Variable("r"), Constant(I["hotspot"]), Constant(I["hot"])))
    sds = Sds()
    sds.add(WindowedTriple("wTemp", (s1, I["temp"], I["hot"]), 2), 5)
    sds.add(WindowedTriple("wLoc", (s1, I["loc"], r1), 0), 100)
    assert (r1, I["hotspot"], I["hot"]) in naive_sds_plus(sds, [rule], db, 6)
    assert (r1, I["hotspot"], I["hot"]) not in naive_sds_plus(sds, [rule], db, 8)


def test_map_fact_survives_sensor_expiry():
    """The long-lived location (map) fact itself stays alive after the
    sensor reading expires."""
    db, I = _setup()
    s1, r1 = db.dictionary.encode("s1"), db.dictionary.encode("r1")
    sds = Sds()
    sds.add(WindowedTriple("wTemp", (s1, I["temp"], I["hot"]), 2), 5)
    sds.add(WindowedTriple("wLoc", (s1, I["loc"], r1), 0), 100)
    alive = sds.alive_facts(50)
    assert (s1, I["loc"], r1) in alive
    assert (s1, I["temp"], I["hot"]) not in alive


def test_incremental_rederives_after_refresh():
    """A fresh sensor reading after expiry re-derives the hotspot with the
    NEW expiry."""
    db, I = _setup()
    s1, r1 = db.dictionary.encode("s1"), db.dictionary.encode("r1")
    rule = _rule(
        [TriplePattern(Variable("s"), Constant(I["temp"]), Constant(I["hot"])),
         TriplePattern(Variable("s"), Constant(I["loc"]), Variable("r"))],
        TriplePattern(Variable("r"), Constant(I["hotspot"]), Constant(I["hot"])))
    sds = Sds()
    sds.add(WindowedTriple("wLoc", (s1, I["loc"], r1), 0), 1000)
    sds.add(WindowedTriple("wTemp", (s1, I["temp"], I["hot"]), 2), 5)
    assert (r1, I["hotspot"], I["hot"]) not in incremental_sds_plus(
        sds, [rule], db, 20)
    sds.add(WindowedTriple("wTemp", (s1, I["temp"], I["hot"]), 21), 5)
    assert (r1, I["hotspot"], I["hot"]) in incremental_sds_plus(
        sds, [rule], db, 24)
    assert (r1, I["hotspot"], I["hot"]) not in incremental_sds_plus(
        sds, [rule], db, 27)


def test_expiry_chain_propagation():
    """2-hop chain: reach expiry = min of the two edges' expiries; the
    3-hop continuation inherits the tightest bound along the chain."""
    db, I = _setup()
    a, b, c, d = (db.dictionary.encode(x) for x in "abcd")
    p = db.dictionary.encode("edge")
    rule1 = _rule([TriplePattern(Variable("x"), Constant(p), Variable("y"))],
                  TriplePattern(Variable("x"), Constant(I["reach"]), Variable("y")))
    rule2 = _rule(
        [TriplePattern(Variable("x"), Constant(I["reach"]), Variable("y")),
         TriplePattern(Variable("y"), Constant(p), Variable("z"))],
        TriplePattern(Variable("x"), Constant(I["reach"]), Variable("z")))
    sds = Sds()
    sds.add(WindowedTriple("w", (a, p, b), 0), 100)   # expires 100
    sds.add(WindowedTriple("w", (b, p, c), 0), 10)    # expires 10
    sds.add(WindowedTriple("w", (c, p, d), 0), 100)   # expires 100
    rules = [rule1, rule2]
    at5 = naive_sds_plus(sds, rules, db, 5)
    assert (a, I["reach"], d) in at5
    at50 = naive_sds_plus(sds, rules, db, 50)
    # the b->c link expired at 10: everything through it is gone
    assert (a, I["reach"], c) not in at50
    assert (a, I["reach"], d) not in at50
    assert (a, I["reach"], b) in at50 and (c, I["reach"], d) in at50
    for ts in (5, 15, 50, 101):
        assert naive_sds_plus(sds, rules, db, ts) == \
            incremental_sds_plus(sds, rules, db, ts), ts


def test_n3_parser_shared_prefixes_missing_dot_and_leftover():
    """ref cross_window_tests.rs parser trio: shared prefixes apply to all
    rules, a missing final `.` is tolerated, leftover input is rejected."""
    import pytest
    from kolibrie_amd.reasoning.n3_rules import parse_n3_rules
    db = SparqlDatabase()
    rules = parse_n3_rules("""
        @prefix ex: <http://e/> .
        { ?x ex:p ?y } => { ?x ex:q ?y } .
        { ?x ex:q ?y . ?y ex:q ?z } => { ?x ex:r ?z }
    """, db)
    assert len(rules) == 2
    assert rules[1].premise[0].p.id == db.dictionary.encode("http://e/q")
    with pytest.raises(ValueError):
        parse_n3_rules("{ ?x <p> ?y } => { ?x <q> ?y } . garbage here", db)
    with pytest.raises(ValueError):
        parse_n3_rules("junk { ?x <p> ?y } => { ?x <q> ?y } .", db)
