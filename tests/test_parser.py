"""Grammar coverage (mirrors kolibrie/tests/parser_test.rs, 45 tests)."""
import pytest

from kolibrie_amd.parsing.sparql import (
    ParseError, parse_combined_query, parse_sparql_query,
)
from kolibrie_amd.parsing import ast as A


def test_simple_select():
    q = parse_sparql_query(
        "SELECT ?s ?o WHERE { ?s <http://e/p> ?o . }")
    assert [p.var for p in q.variables] == ["s", "o"]
    assert isinstance(q.where, A.GBgp)
    assert q.where.patterns[0].p == "<http://e/p>"


def test_prefixes_and_prefixed_names():
    cq = parse_combined_query("""
        PREFIX foaf: <http://xmlns.com/foaf/0.1/>
        SELECT ?n WHERE { ?x foaf:name ?n }
    """)
    assert cq.prefixes["foaf"] == "http://xmlns.com/foaf/0.1/"
    assert cq.select.where.patterns[0].p == "foaf:name"


def test_select_star_distinct_limit():
    q = parse_sparql_query(
        "SELECT DISTINCT * WHERE { ?s ?p ?o } LIMIT 10 OFFSET 5")
    assert q.select_star and q.distinct
    assert q.limit == 10 and q.offset == 5


def test_semicolon_comma_groups():
    q = parse_sparql_query(
        'SELECT * WHERE { ?s <http://e/a> ?x ; <http://e/b> ?y , ?z . }')
    pats = q.where.patterns
    assert len(pats) == 3
    assert all(p.s == "?s" for p in pats)


def test_filter_expression():
    q = parse_sparql_query(
        "SELECT ?s WHERE { ?s <http://e/age> ?a . FILTER(?a > 30 && ?a < 50) }")
    assert isinstance(q.where, A.GFilter)
    assert isinstance(q.where.expr, A.EAnd)


def test_filter_scope_deferred_to_group_end():
    q = parse_sparql_query(
        "SELECT ?s WHERE { FILTER(?a > 1) ?s <http://e/p> ?a . }")
    # filter wraps the whole group even when written first
    assert isinstance(q.where, A.GFilter)
    assert isinstance(q.where.inner, A.GBgp)


def test_union():
    q = parse_sparql_query(
        "SELECT ?x WHERE { { ?x <http://e/a> ?y } UNION { ?x <http://e/b> ?y } }")
    assert isinstance(q.where, A.GUnion)


def test_graph_pattern():
    q = parse_sparql_query(
        "SELECT ?x WHERE { GRAPH <http://e/g> { ?x <http://e/p> ?y } }")
    assert isinstance(q.where, A.GGraph)
    q2 = parse_sparql_query(
        "SELECT ?g WHERE { GRAPH ?g { ?x <http://e/p> ?y } }")
    assert q2.where.graph == "?g"


def test_bind():
    q = parse_sparql_query(
        'SELECT ?c WHERE { ?x <http://e/p> ?y . BIND(CONCAT(?y, "!") AS ?c) }')
    assert isinstance(q.where, A.GBind)
    assert q.where.var == "c"
    assert q.where.expr.name == "CONCAT"


def test_values_single_and_multi():
    q = parse_sparql_query(
        'SELECT ?x WHERE { VALUES ?x { <http://e/a> <http://e/b> } }')
    assert isinstance(q.where, A.GValues)
    assert len(q.where.rows) == 2
    q2 = parse_sparql_query(
        'SELECT * WHERE { VALUES (?x ?y) { (<http://e/a> UNDEF) ("1" "2") } }')
    assert q2.where.variables == ["x", "y"]
    assert q2.where.rows[0][1] is None


def test_subquery():
    q = parse_sparql_query("""
        SELECT ?s WHERE {
          { SELECT ?s WHERE { ?s <http://e/p> ?o } LIMIT 2 }
        }""")
    assert isinstance(q.where, A.GSubQuery)
    assert q.where.select.limit == 2


def test_aggregates():
    q = parse_sparql_query(
        "SELECT (COUNT(?x) AS ?c) (SUM(?v) AS ?s) WHERE { ?x <http://e/v> ?v } GROUP BY ?x")
    assert q.variables[0].aggregate == "COUNT"
    assert q.variables[1].alias == "s"
    assert q.group_by == ["x"]


def test_order_by():
    q = parse_sparql_query(
        "SELECT ?s WHERE { ?s <http://e/p> ?o } ORDER BY DESC(?o) ?s")
    assert q.order_by[0].descending
    assert q.order_by[1].var == "s"


def test_from_and_from_named():
    q = parse_sparql_query(
        "SELECT ?s FROM <http://e/g1> FROM NAMED <http://e/g2> WHERE { ?s ?p ?o }")
    assert q.from_graphs == ["<http://e/g1>"]
    assert q.from_named == ["<http://e/g2>"]


def test_quoted_triple_pattern():
    q = parse_sparql_query(
        "SELECT ?c WHERE { << ?s <http://e/p> ?o >> <http://e/certainty> ?c }")
    assert q.where.patterns[0].s.startswith("<<")


def test_insert_data():
    cq = parse_combined_query(
        'INSERT DATA { <http://e/a> <http://e/p> "v" . }')
    assert cq.updates[0].kind == "insert_data"
    assert cq.updates[0].quads[0].o == '"v"'


def test_insert_data_named_graph():
    cq = parse_combined_query(
        'INSERT DATA { GRAPH <http://e/g> { <http://e/a> <http://e/p> "v" } }')
    assert cq.updates[0].quads[0].g == "<http://e/g>"


def test_delete_insert_where():
    cq = parse_combined_query("""
        DELETE { ?s <http://e/old> ?o }
        INSERT { ?s <http://e/new> ?o }
        WHERE { ?s <http://e/old> ?o }""")
    op = cq.updates[0]
    assert op.kind == "modify"
    assert op.delete_templates and op.insert_templates and op.where


def test_clear_create_drop():
    cq = parse_combined_query("CREATE GRAPH <http://e/g> ; CLEAR GRAPH <http://e/g> ; DROP SILENT GRAPH <http://e/g>")
    kinds = [u.kind for u in cq.updates]
    assert kinds == ["create", "clear", "drop"]
    assert cq.updates[2].silent


def test_register_rsp_query():
    cq = parse_combined_query("""
        REGISTER RSTREAM <http://out> AS
        SELECT ?a
        FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE PT10S STEP PT5S]
        WHERE { WINDOW <http://w1> { ?a <http://e/p> ?b } }
    """)
    r = cq.register
    assert r.stream_type == "RSTREAM"
    assert r.windows[0].spec.width == 10
    assert r.windows[0].spec.slide == 5
    assert isinstance(r.select.where, A.GWindowBlock)


def test_window_spec_variants():
    cq = parse_combined_query("""
        REGISTER ISTREAM <http://out> AS SELECT ?a
        FROM NAMED WINDOW :w ON :s [TUMBLING 30 REPORT ON_WINDOW_CLOSE TICK TIME_DRIVEN]
        WHERE { WINDOW :w { ?a ?p ?b } }
    """)
    spec = cq.register.windows[0].spec
    assert spec.window_type == "TUMBLING"
    assert spec.width == 30 and spec.slide == 30
    assert spec.report == "ON_WINDOW_CLOSE"
    assert spec.tick == "TIME_DRIVEN"


def test_window_policy():
    cq = parse_combined_query("""
        REGISTER RSTREAM <http://out> AS SELECT ?a
        FROM NAMED WINDOW :w ON :s [RANGE PT5S] WITH POLICY TIMEOUT PT2S
        WHERE { WINDOW :w { ?a ?p ?b } }
    """)
    pol = cq.register.windows[0].policy
    assert pol.kind == "Timeout" and pol.timeout_ms == 2000


def test_rule_basic():
    cq = parse_combined_query("""
        RULE :Grandparent :- CONSTRUCT { ?x <http://e/grandparent> ?z }
        WHERE { ?x <http://e/parent> ?y . ?y <http://e/parent> ?z } .
    """)
    r = cq.rules[0]
    assert r.name == ":Grandparent"
    assert len(r.conclusions) == 1
    assert isinstance(r.body, A.GBgp)
    assert len(r.body.patterns) == 2


def test_rule_with_prob():
    cq = parse_combined_query("""
        RULE :Risk PROB(provenance=minmax, threshold=0.5, confidence=0.9) :-
        CONSTRUCT { ?x <http://e/risky> "yes" } WHERE { ?x <http://e/score> ?s } .
    """)
    r = cq.rules[0]
    assert r.prob.provenance == "minmax"
    assert r.prob.threshold == 0.5
    assert r.prob.confidence == 0.9


def test_rule_with_not():
    cq = parse_combined_query("""
        RULE :OnlyNew :- CONSTRUCT { ?x <http://e/new> "1" }
        WHERE { ?x <http://e/item> ?y . NOT { ?x <http://e/old> ?y } } .
    """)
    r = cq.rules[0]
    assert len(r.negated) == 1
    assert r.negated[0].p == "<http://e/old>"


def test_rule_with_window():
    cq = parse_combined_query("""
        RULE :Alert :- RSTREAM FROM NAMED WINDOW :w ON :s [RANGE PT10S]
        CONSTRUCT { ?m <http://e/alert> "hot" }
        WHERE { ?m <http://e/temp> ?t . FILTER(?t > 90) } .
    """)
    r = cq.rules[0]
    assert r.stream_type == "RSTREAM"
    assert len(r.windows) == 1


def test_model_decl():
    cq = parse_combined_query("""
        MODEL "fraud" { ARCH MLP { HIDDEN [64, 32] } OUTPUT BINARY {"fraud"} }
        SELECT ?x WHERE { ?x ?p ?o }
    """)
    m = cq.models[0]
    assert m.name == "fraud"
    assert m.options["hidden"] == "64,32"
    assert m.options["output"] == "BINARY"


def test_neural_relation_decl():
    cq = parse_combined_query("""
        NEURAL RELATION <http://e/suspicious> USING MODEL "fraud" {
            INPUT { ?x <http://e/amount> ?a }
            FEATURES { ?a }
        }
        SELECT ?x WHERE { ?x ?p ?o }
    """)
    nr = cq.neural_relations[0]
    assert nr.model == "fraud"
    assert nr.options["features"] == "a"


def test_parse_error_reporting():
    with pytest.raises(ParseError) as e:
        parse_sparql_query("SELECT ?x WHERE { ?x <http://e/p }")
    assert "line" in str(e.value)


def test_retrieve_clause():
    cq = parse_combined_query("RETRIEVE SOME LATENT <http://e/s1> <http://e/s2>")
    assert cq.retrieve.mode == "SOME"
    assert len(cq.retrieve.streams) == 2


# ---- grammar-surface parity probes (ref parser_test.rs) ----

def _ok(q):
    from kolibrie_amd.parsing.sparql import parse_combined_query
    return parse_combined_query(q)


def test_a_syntax_in_select_and_rules():
    _ok('SELECT ?x WHERE { ?x a <http://e/Person> }')
    cq = _ok('RULE :t :- CONSTRUCT { ?x a <http://e/B> } '
             'WHERE { ?x a <http://e/A> } '
             'SELECT ?x WHERE { ?x a <http://e/B> }')
    assert cq.rules


def test_case_insensitive_keywords_nested_graph_union():
    cq = _ok('select ?x where { { ?x a <http://e/A> } union '
             '{ graph <http://g> { ?x <http://e/p> ?y } } }')
    assert cq.select is not None


def test_comments_inside_group():
    _ok('SELECT ?x WHERE { ?x <http://e/p> ?y . # a comment\n'
        ' FILTER(?y > 3) }')


def test_literal_escapes_and_language_tags():
    _ok('SELECT ?x WHERE { ?x <http://e/label> "hi"@en }')
    _ok('SELECT ?x WHERE { ?x <http://e/p> "line\\nbreak \\"q\\"" }')


def test_prefixed_names_with_dots():
    _ok('PREFIX ex: <http://e/> SELECT ?x WHERE { ?x ex:foo.bar ?y }')


def test_prob_annotation_variants():
    base = (' :- CONSTRUCT { ?x <http://e/q> ?y } '
            'WHERE { ?x <http://e/p> ?y } SELECT ?x WHERE { ?x <http://e/q> ?y }')
    for ann in ("PROB(combination=min)", "PROB(provenance=topk, k=4)",
                "PROB(provenance=wmc)",
                "PROB(provenance=hybrid, threshold=0.7, band_epsilon=0.05)"):
        cq = _ok(f'RULE :r {ann}{base}')
        assert cq.rules[0].prob is not None
    # no annotation still works
    cq = _ok('RULE :r' + base)
    assert cq.rules[0].prob is None


def test_hybrid_prob_validation():
    """ref parser_test.rs hybrid_probability_annotation_rejects_*: hybrid
    requires a threshold, rejects unknown/duplicate keys + confidence,
    rejects invalid auto policies; auto:cost(fp,fn) -> CostRatio."""
    import pytest
    from kolibrie_amd.parsing.sparql import ParseError
    base = (' :- CONSTRUCT {{ ?x <http://e/q> ?y }} '
            'WHERE {{ ?x <http://e/p> ?y }} '
            'SELECT ?x WHERE {{ ?x <http://e/q> ?y }}')

    def rule(ann):
        return _ok(('RULE :r ' + ann + base).format())

    cq = rule('PROB(provenance=hybrid, threshold=auto:cost(fp=1,fn=3))')
    pa = cq.rules[0].prob
    assert abs(pa.threshold - 0.25) < 1e-9
    assert pa.extra["threshold_policy"] == "CostRatio"

    invalid = [
        'PROB(provenance=hybrid)',                                # no threshold
        'PROB(provenance=hybrid, threshold=0.7, mystery=1)',      # unknown key
        'PROB(provenance=hybrid, threshold=0.7, confidence=0.9)', # confidence
        'PROB(provenance=hybrid, threshold=auto:quantile(0.9))',  # quantile
        'PROB(provenance=hybrid, threshold=auto:cost(fp=1))',     # missing fn
        'PROB(provenance=hybrid, threshold=auto:cost(fp=0,fn=0))',# zero total
        'PROB(provenance=hybrid, threshold=auto:cost(fp=-1,fn=2))',
        'PROB(provenance=hybrid, threshold=1.7)',                 # out of range
        'PROB(provenance=hybrid, threshold=0.4, threshold=0.6)',  # duplicate
    ]
    for ann in invalid:
        with pytest.raises(ParseError):
            rule(ann)
    # non-hybrid still tolerates extra keys
    cq = rule('PROB(provenance=topk, k=4, mystery=2)')
    assert cq.rules[0].prob.provenance == "topk"
