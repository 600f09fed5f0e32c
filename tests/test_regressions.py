"""Dataset / update regressions (mirrors sparql_dataset_regressions_test.rs,
named_graph_test.rs, hybrid_test.rs shapes) + exec-stats counters."""
import pytest

from kolibrie_amd import SparqlDatabase

EX = "http://example.org/"


def test_nquads_roundtrip_with_graphs():
    db = SparqlDatabase()
    db.query(f"""INSERT DATA {{
        <{EX}a> <{EX}p> "v0" .
        GRAPH <{EX}g1> {{ <{EX}b> <{EX}p> "v1" }}
        GRAPH <{EX}g2> {{ <{EX}c> <{EX}p> "v2" }}
    }}""")
    text = db.generate_nquads()
    db2 = SparqlDatabase()
    db2.parse_nquads(text)
    assert db2.generate_nquads() == text
    rows = db2.query(f"SELECT ?g WHERE {{ GRAPH ?g {{ ?s <{EX}p> ?o }} }}")
    assert sorted(r[0] for r in rows) == [f"{EX}g1", f"{EX}g2"]


def test_illegal_update_positions():
    db = SparqlDatabase()
    with pytest.raises(ValueError):
        db.query(f'INSERT DATA {{ "literal" <{EX}p> "v" }}')
    with pytest.raises(ValueError):
        db.query(f'INSERT DATA {{ <{EX}s> "litpred" "v" }}')
    with pytest.raises(ValueError):
        db.query(f'INSERT DATA {{ ?var <{EX}p> "v" }}')


def test_create_existing_graph_errors_unless_silent():
    db = SparqlDatabase()
    db.query(f"CREATE GRAPH <{EX}g>")
    with pytest.raises(ValueError):
        db.query(f"CREATE GRAPH <{EX}g>")
    db.query(f"CREATE SILENT GRAPH <{EX}g>")   # no error
    db.query(f"DROP GRAPH <{EX}g>")
    with pytest.raises(ValueError):
        db.query(f"DROP GRAPH <{EX}g>")


def test_union_multiplicity_preserved():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", '"v"')
    rows = db.query(f"""SELECT ?s WHERE {{
        {{ ?s <{EX}p> "v" }} UNION {{ ?s <{EX}p> "v" }} UNION {{ ?s <{EX}p> "v" }}
    }}""")
    assert len(rows) == 3


def test_values_multiplicity_with_undef():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", '"1"')
    rows = db.query(f"""SELECT ?x ?y WHERE {{
        VALUES (?x ?y) {{ (<{EX}a> "tag1") (<{EX}a> UNDEF) }}
        ?x <{EX}p> ?o .
    }}""")
    assert len(rows) == 2


def test_exec_stats_counters():
    from kolibrie_amd.engine import exec_stats
    db = SparqlDatabase()
    for i in range(10):
        db.add_triple(f"<{EX}e{i}>", f"<{EX}p>", f'"{i}"')
    exec_stats.reset()
    db.query(f"SELECT ?s WHERE {{ ?s <{EX}p> ?o }}")
    snap = exec_stats.snapshot()
    assert snap["SCAN_PROBES"] >= 1
    assert snap["ROWS_EMITTED"] >= 10
    exec_stats.reset()
    assert exec_stats.snapshot()["ROWS_EMITTED"] == 0


def test_hybrid_recursive_rule_rejected():
    # hybrid evaluation requires monotone rules; recursion through negation
    # is rejected (ref hybrid_test.rs recursion rejection)
    from kolibrie_amd.reasoning.hybrid import validate_monotone
    from kolibrie_amd.reasoning.rule import Rule
    from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable
    bad = Rule(
        premise=[TriplePattern(Variable("x"), Constant(1), Variable("y"))],
        negative_premise=[TriplePattern(Variable("x"), Constant(2), Variable("y"))],
        conclusion=[TriplePattern(Variable("x"), Constant(2), Variable("y"))],
    )
    with pytest.raises(ValueError):
        validate_monotone([bad])


def test_typed_hybrid_status_annotations():
    from kolibrie_amd.reasoning.hybrid import (
        HybridConfig, LineageStore, evaluate_hybrid, encode_results_rdf_star,
        HybridProbabilityResult,
    )
    db = SparqlDatabase()
    s = db.dictionary.encode("s")
    p = db.dictionary.encode("p")
    o = db.dictionary.encode("o")
    st = LineageStore()
    res = evaluate_hybrid(st, st.leaf(1), {1: 0.93}, HybridConfig())
    encode_results_rdf_star({(s, p, o): res}, db)
    rows = db.query("""
        SELECT ?st WHERE {
            ?t <http://kolibrie.amd/hybrid#status> ?st . FILTER(isTRIPLE(?t))
        }""")
    assert rows and rows[0][0] in ("Decided", "DecidedExact")


def test_count_star_fast_path_matches_generic():
    """The colless COUNT(*) fast path must agree with the generic
    finalize path (cached vs uncached entry), incl. empty data and
    LIMIT/OFFSET edges."""
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.engine.query import execute_query, execute_select
    from kolibrie_amd.parsing.sparql import parse_combined_query
    db = SparqlDatabase()
    for i in range(20):
        db.add_triple(f"<http://e/s{i}>", "<http://e/p>", f"<http://e/o{i%3}>")
        db.add_triple(f"<http://e/s{i}>", "<http://e/q>", "<http://e/z>")
    q = ('SELECT (COUNT(*) AS ?c) WHERE { ?s <http://e/p> ?o . '
         '?s <http://e/q> ?z }')
    cached = execute_query(q, db)
    cq = parse_combined_query(q)
    plain = execute_select(cq.select, db, dict(db.prefixes))
    assert cached == plain == [["20"]]
    assert execute_query(q + " LIMIT 0", db) == []
    assert execute_query(q + " OFFSET 1", db) == []
    assert execute_query(q, SparqlDatabase()) == [["0"]]


def test_scan_unit_extreme_ids():
    """Range-bound arithmetic edges: max-positive i32 leading constant
    (k+1 overflow guard) and negative (quoted-style) ids."""
    import torch
    from kolibrie_amd.storage.dataset import GraphIndex
    from kolibrie_amd.engine.scan import scan_unit
    s = torch.tensor([0x7FFFFFFF, 0x7FFFFFFF, 5],
                     dtype=torch.int64).to(torch.int32)
    p = torch.tensor([1, 2, 1], dtype=torch.int32)
    o = torch.tensor([10, 11, 12], dtype=torch.int32)
    gi = GraphIndex.from_columns(s, p, o, device="cpu")
    _, _, ro = scan_unit(gi, {0: 0x7FFFFFFF})
    assert sorted(ro.tolist()) == [10, 11]
    _, _, ro = scan_unit(gi, {0: 0x7FFFFFFF, 1: 2})
    assert ro.tolist() == [11]
    s2 = torch.tensor([-5, -5, 3], dtype=torch.int32)
    gi2 = GraphIndex.from_columns(s2, p, o, device="cpu")
    _, _, ro = scan_unit(gi2, {0: -5})
    assert sorted(ro.tolist()) == [10, 11]


def test_update_bnode_allocation_skips_lexical_collisions():
    """Pre-existing '_:updN' labels must not alias fresh update bnodes
    (ref sparql_dataset_regressions_test.rs)."""
    from kolibrie_amd import SparqlDatabase
    db = SparqlDatabase()
    db.add_triple("_:upd1", "<http://e/x>", '"old"')     # collide on purpose
    db.add_triple("<http://e/s>", "<http://e/p>", '"v"')
    db.query('INSERT { ?s <http://e/tag> _:b } WHERE { ?s <http://e/p> ?o }')
    tags = db.query('SELECT ?b WHERE { <http://e/s> <http://e/tag> ?b }')
    assert len(tags) == 1
    # the fresh bnode must NOT be the pre-existing _:upd1
    old = db.query('SELECT ?o WHERE { _:upd1 <http://e/x> ?o }')
    assert old == [["old"]]
    assert tags[0][0] != "_:upd1"


def test_rebuild_indexes_keeps_named_quads_and_empty_graphs():
    from kolibrie_amd import SparqlDatabase
    db = SparqlDatabase()
    db.query('INSERT DATA { GRAPH <http://g1> { <http://e/a> <http://e/p> <http://e/b> } }')
    db.query('CREATE GRAPH <http://gEmpty>')
    db.build_all_indexes()
    gids = {db.decode_term(g) for g in db.store.named_graph_ids()}
    assert "http://g1" in gids and "http://gEmpty" in gids
    assert db.query('SELECT ?s WHERE { GRAPH <http://g1> { ?s ?p ?o } }') == \
        [["http://e/a"]]


def test_out_of_vocabulary_ids_value_zero():
    """Ids beyond the interned vocabulary (bulk synthetic loads) must
    evaluate to 0.0 in FILTER/ORDER/aggregates — matching the K5 kernel's
    bound check, NOT the last dictionary entry's value (regression found
    by the CPU-vs-GPU differential sweep, seed 404)."""
    import torch
    from kolibrie_amd import SparqlDatabase
    db = SparqlDatabase()
    db.dictionary.encode("999999")  # a BIG numeric literal as last entry
    p = db.dictionary.encode("http://e/v")
    s = torch.tensor([5_000_000, 5_000_001], dtype=torch.int32)
    db.store.insert_bulk(0, s, torch.full((2,), p, dtype=torch.int32),
                         s + 10_000)  # objects also out-of-vocab
    assert db.query(
        'SELECT (COUNT(*) AS ?c) WHERE { ?s <http://e/v> ?o . '
        'FILTER(?o > 1) }') == [["0"]]
    assert db.query(
        'SELECT (COUNT(*) AS ?c) WHERE { ?s <http://e/v> ?o . '
        'FILTER(?o < 1) }') == [["2"]]
    assert db.query(
        'SELECT (SUM(?o) AS ?t) WHERE { ?s <http://e/v> ?o }') == [["0"]]


def test_rdf_parsers_raise_clean_errors_on_truncation():
    """Truncated / mutated RDF inputs raise ValueError, never IndexError
    (fuzz-found)."""
    import pytest
    from kolibrie_amd import SparqlDatabase
    for text in ["<http://e/", "<< <http://e/a> <http://e/b> <http://e/",
                 "<http://e/s> <http://e/p>"]:
        for fn in ("parse_ntriples", "parse_turtle", "parse_nquads",
                   "parse_n3"):
            db = SparqlDatabase()
            try:
                getattr(db, fn)(text)
            except ValueError:
                pass  # clean parse error (or tolerated partial) is fine


def test_execution_fuzz_deterministic():
    """200 random (valid) query shapes execute without internal errors
    (ValueError for user-level issues is acceptable)."""
    import random
    from kolibrie_amd import SparqlDatabase
    rng = random.Random(7)
    db = SparqlDatabase()
    for i in range(120):
        db.add_triple(f"<http://e/s{i%20}>", f"<http://e/p{i%4}>", f'"{i%9}"')
        db.add_triple(f"<http://e/s{i%20}>", "<http://e/link>",
                      f"<http://e/s{(i*7)%20}>")
    VARS = ["?a", "?b", "?c"]

    def pat():
        s = rng.choice(VARS + ["<http://e/s3>"])
        p = rng.choice([f"<http://e/p{rng.randrange(4)}>",
                        "<http://e/link>", "?pp"])
        o = rng.choice(VARS + ['"5"', "<http://e/s7>"])
        return f"{s} {p} {o}"

    for _ in range(200):
        pats = " . ".join(pat() for _ in range(rng.randrange(1, 4)))
        extra = ""
        if rng.random() < 0.4:
            extra += (f" FILTER({rng.choice(VARS)} "
                      f"{rng.choice(['>', '<', '=', '!='])} "
                      f"{rng.randrange(15)})")
        if rng.random() < 0.3:
            extra += f" OPTIONAL {{ {pat()} }}"
        if rng.random() < 0.2:
            extra += f" MINUS {{ {pat()} }}"
        proj = rng.choice(["*", "?a", "?a ?b", "(COUNT(*) AS ?n)",
                           "DISTINCT ?a"])
        mods = ""
        if rng.random() < 0.3:
            mods += f" ORDER BY {rng.choice(VARS)}"
        if rng.random() < 0.3:
            mods += f" LIMIT {rng.randrange(5)}"
        q = f"SELECT {proj} WHERE {{ {pats}{extra} }}{mods}"
        try:
            db.query(q)
        except ValueError:
            pass
