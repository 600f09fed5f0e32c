"""End-to-end SELECT/UPDATE queries (mirrors kolibrie/tests/
sparql_unified_test.rs, sparql_graph_test.rs, integration_test.rs)."""
import pytest

from kolibrie_amd import SparqlDatabase

EX = "http://example.org/"


def _people_db():
    db = SparqlDatabase()
    data = [
        ("alice", "name", '"Alice"'),
        ("alice", "age", '"30"'),
        ("alice", "worksFor", "<http://example.org/acme>"),
        ("bob", "name", '"Bob"'),
        ("bob", "age", '"45"'),
        ("bob", "worksFor", "<http://example.org/acme>"),
        ("carol", "name", '"Carol"'),
        ("carol", "age", '"22"'),
        ("carol", "worksFor", "<http://example.org/initech>"),
    ]
    for s, p, o in data:
        db.add_triple(f"<{EX}{s}>", f"<{EX}{p}>", o)
    return db


def test_single_pattern(db):
    db.add_triple(f"<{EX}a>", f"<{EX}p>", '"v1"')
    db.add_triple(f"<{EX}b>", f"<{EX}p>", '"v2"')
    rows = db.query(f"SELECT ?s ?o WHERE {{ ?s <{EX}p> ?o }}")
    assert sorted(rows) == [[f"{EX}a", "v1"], [f"{EX}b", "v2"]]


def test_two_pattern_join():
    db = _people_db()
    rows = db.query(f"""
        SELECT ?n ?a WHERE {{
            ?x <{EX}name> ?n .
            ?x <{EX}age> ?a .
        }}""")
    assert sorted(rows) == [["Alice", "30"], ["Bob", "45"], ["Carol", "22"]]


def test_three_pattern_star_join():
    db = _people_db()
    rows = db.query(f"""
        SELECT ?n ?a ?w WHERE {{
            ?x <{EX}name> ?n .
            ?x <{EX}age> ?a .
            ?x <{EX}worksFor> ?w .
        }}""")
    assert len(rows) == 3
    assert ["Alice", "30", f"{EX}acme"] in rows


def test_filter_numeric():
    db = _people_db()
    rows = db.query(f"""
        SELECT ?n WHERE {{
            ?x <{EX}name> ?n . ?x <{EX}age> ?a .
            FILTER(?a > 25 && ?a <= 45)
        }}""")
    assert sorted(r[0] for r in rows) == ["Alice", "Bob"]


def test_filter_string_equality():
    db = _people_db()
    rows = db.query(f"""
        SELECT ?x WHERE {{ ?x <{EX}name> ?n . FILTER(?n = "Alice") }}""")
    assert rows == [[f"{EX}alice"]]


def test_filter_var_var_id_equality():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}x>", f"<{EX}a>", '"same"')
    db.add_triple(f"<{EX}x>", f"<{EX}b>", '"same"')
    db.add_triple(f"<{EX}y>", f"<{EX}a>", '"one"')
    db.add_triple(f"<{EX}y>", f"<{EX}b>", '"two"')
    rows = db.query(f"""
        SELECT ?s WHERE {{ ?s <{EX}a> ?v1 . ?s <{EX}b> ?v2 . FILTER(?v1 = ?v2) }}""")
    assert rows == [[f"{EX}x"]]


def test_union_multiplicity():
    # UNION preserves duplicates (multiset semantics)
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", '"v"')
    rows = db.query(f"""
        SELECT ?s WHERE {{
            {{ ?s <{EX}p> ?o }} UNION {{ ?s <{EX}p> ?o }}
        }}""")
    assert len(rows) == 2


def test_bind_concat():
    db = _people_db()
    rows = db.query(f"""
        SELECT ?g WHERE {{
            ?x <{EX}name> ?n . BIND(CONCAT("Hello ", ?n) AS ?g)
        }}""")
    assert ["Hello Alice"] in rows


def test_values_inline():
    db = _people_db()
    rows = db.query(f"""
        SELECT ?n WHERE {{
            VALUES ?x {{ <{EX}alice> <{EX}bob> }}
            ?x <{EX}name> ?n .
        }}""")
    assert sorted(r[0] for r in rows) == ["Alice", "Bob"]


def test_values_undef_wildcard():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", '"1"')
    db.add_triple(f"<{EX}b>", f"<{EX}p>", '"2"')
    rows = db.query(f"""
        SELECT ?x ?o WHERE {{
            VALUES (?x ?o) {{ (<{EX}a> UNDEF) }}
            ?x <{EX}p> ?o .
        }}""")
    assert rows == [[f"{EX}a", "1"]]


def test_distinct_order_limit():
    db = _people_db()
    rows = db.query(f"""
        SELECT DISTINCT ?w WHERE {{ ?x <{EX}worksFor> ?w }} ORDER BY ?w""")
    assert rows == [[f"{EX}acme"], [f"{EX}initech"]]
    rows = db.query(f"""
        SELECT ?a WHERE {{ ?x <{EX}age> ?a }} ORDER BY DESC(?a) LIMIT 2""")
    assert rows == [["45"], ["30"]]


def test_aggregates_group_by():
    db = _people_db()
    rows = db.query(f"""
        SELECT ?w (COUNT(?x) AS ?c) WHERE {{
            ?x <{EX}worksFor> ?w .
        }} GROUP BY ?w ORDER BY ?w""")
    assert rows == [[f"{EX}acme", "2"], [f"{EX}initech", "1"]]


def test_aggregate_sum_avg():
    db = _people_db()
    rows = db.query(f"""
        SELECT (SUM(?a) AS ?s) (AVG(?a) AS ?m) (MIN(?a) AS ?lo) (MAX(?a) AS ?hi)
        WHERE {{ ?x <{EX}age> ?a }}""")
    assert rows == [["97", repr(97 / 3), "22", "45"]]


def test_subquery_with_limit():
    db = _people_db()
    rows = db.query(f"""
        SELECT ?n WHERE {{
            {{ SELECT ?x WHERE {{ ?x <{EX}age> ?a }} ORDER BY DESC(?a) LIMIT 1 }}
            ?x <{EX}name> ?n .
        }}""")
    assert rows == [["Bob"]]


def test_named_graph_query():
    db = SparqlDatabase()
    db.add_quad_parts(f"<{EX}a>", f"<{EX}p>", '"in-g1"', f"<{EX}g1>")
    db.add_quad_parts(f"<{EX}b>", f"<{EX}p>", '"in-g2"', f"<{EX}g2>")
    db.add_triple(f"<{EX}c>", f"<{EX}p>", '"default"')
    rows = db.query(f"SELECT ?o WHERE {{ GRAPH <{EX}g1> {{ ?s <{EX}p> ?o }} }}")
    assert rows == [["in-g1"]]
    # GRAPH ?g never matches the default graph
    rows = db.query(f"SELECT ?g ?o WHERE {{ GRAPH ?g {{ ?s <{EX}p> ?o }} }}")
    assert sorted(rows) == [[f"{EX}g1", "in-g1"], [f"{EX}g2", "in-g2"]]


def test_graph_var_join_consistency():
    db = SparqlDatabase()
    db.add_quad_parts(f"<{EX}a>", f"<{EX}p>", f"<{EX}b>", f"<{EX}g1>")
    db.add_quad_parts(f"<{EX}b>", f"<{EX}q>", '"x"', f"<{EX}g1>")
    db.add_quad_parts(f"<{EX}b>", f"<{EX}q>", '"y"', f"<{EX}g2>")
    rows = db.query(f"""
        SELECT ?o WHERE {{
            GRAPH ?g {{ ?a <{EX}p> ?b . ?b <{EX}q> ?o }}
        }}""")
    assert rows == [["x"]]


def test_from_merged_dedup():
    db = SparqlDatabase()
    db.add_quad_parts(f"<{EX}a>", f"<{EX}p>", '"v"', f"<{EX}g1>")
    db.add_quad_parts(f"<{EX}a>", f"<{EX}p>", '"v"', f"<{EX}g2>")
    rows = db.query(f"""
        SELECT ?s FROM <{EX}g1> FROM <{EX}g2> WHERE {{ ?s <{EX}p> "v" }}""")
    assert len(rows) == 1  # merged-FROM default dedups


def test_insert_data_and_delete():
    db = SparqlDatabase()
    db.query(f'INSERT DATA {{ <{EX}a> <{EX}p> "v1" . <{EX}b> <{EX}p> "v2" }}')
    assert db.triple_count() == 2
    db.query(f'DELETE DATA {{ <{EX}a> <{EX}p> "v1" }}')
    assert db.triple_count() == 1


def test_modify_update():
    db = _people_db()
    db.query(f"""
        DELETE {{ ?x <{EX}worksFor> <{EX}acme> }}
        INSERT {{ ?x <{EX}worksFor> <{EX}megacorp> }}
        WHERE {{ ?x <{EX}worksFor> <{EX}acme> }}""")
    rows = db.query(f"SELECT ?x WHERE {{ ?x <{EX}worksFor> <{EX}megacorp> }}")
    assert len(rows) == 2
    rows = db.query(f"SELECT ?x WHERE {{ ?x <{EX}worksFor> <{EX}acme> }}")
    assert rows == []


def test_update_in_named_graph():
    db = SparqlDatabase()
    db.query(f'INSERT DATA {{ GRAPH <{EX}g> {{ <{EX}a> <{EX}p> "v" }} }}')
    rows = db.query(f"SELECT ?o WHERE {{ GRAPH <{EX}g> {{ ?s ?p ?o }} }}")
    assert rows == [["v"]]
    db.query(f"CLEAR GRAPH <{EX}g>")
    rows = db.query(f"SELECT ?o WHERE {{ GRAPH <{EX}g> {{ ?s ?p ?o }} }}")
    assert rows == []


def test_rdf_star_query():
    db = SparqlDatabase()
    db.add_triple(f"<< <{EX}s> <{EX}p> <{EX}o> >>", f"<{EX}certainty>", '"0.9"')
    rows = db.query(f"""
        SELECT ?c WHERE {{ << <{EX}s> <{EX}p> <{EX}o> >> <{EX}certainty> ?c }}""")
    assert rows == [["0.9"]]


def test_rdf_star_variable_inside_quoted():
    db = SparqlDatabase()
    db.add_triple(f"<< <{EX}s1> <{EX}p> <{EX}o1> >>", f"<{EX}cert>", '"0.9"')
    db.add_triple(f"<< <{EX}s2> <{EX}p> <{EX}o2> >>", f"<{EX}cert>", '"0.5"')
    rows = db.query(f"""
        SELECT ?s ?c WHERE {{ << ?s <{EX}p> ?o >> <{EX}cert> ?c }}""")
    assert sorted(rows) == [[f"{EX}s1", "0.9"], [f"{EX}s2", "0.5"]]


def test_istriple_builtin():
    db = SparqlDatabase()
    db.add_triple(f"<< <{EX}s> <{EX}p> <{EX}o> >>", f"<{EX}cert>", '"0.9"')
    db.add_triple(f"<{EX}plain>", f"<{EX}cert>", '"1.0"')
    rows = db.query(f"""
        SELECT ?c WHERE {{ ?t <{EX}cert> ?c . FILTER(isTRIPLE(?t)) }}""")
    assert rows == [["0.9"]]


def test_udf_in_bind():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}v>", '"5"')
    db.register_udf("DOUBLE", lambda x: str(int(x) * 2))
    rows = db.query(f"""
        SELECT ?d WHERE {{ ?s <{EX}v> ?v . BIND(DOUBLE(?v) AS ?d) }}""")
    assert rows == [["10"]]


def test_repeated_variable_in_pattern():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", f"<{EX}a>")
    db.add_triple(f"<{EX}a>", f"<{EX}p>", f"<{EX}b>")
    rows = db.query(f"SELECT ?x WHERE {{ ?x <{EX}p> ?x }}")
    assert rows == [[f"{EX}a"]]


def test_query_only_endpoint_rejects_update():
    from kolibrie_amd.engine.query import execute_sparql_query
    db = SparqlDatabase()
    with pytest.raises(ValueError):
        execute_sparql_query(f'INSERT DATA {{ <{EX}a> <{EX}p> "v" }}', db)


def test_prefixed_query():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}alice>", "<http://xmlns.com/foaf/0.1/name>", '"Alice"')
    rows = db.query("""
        PREFIX foaf: <http://xmlns.com/foaf/0.1/>
        SELECT ?n WHERE { ?x foaf:name ?n }""")
    assert rows == [["Alice"]]


def test_optional_basic():
    """OPTIONAL left outer join (engine extension beyond the reference's
    SPARQL subset): unmatched left rows keep UNBOUND right vars."""
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}name>", '"Alice"')
    db.add_triple(f"<{EX}b>", f"<{EX}name>", '"Bob"')
    db.add_triple(f"<{EX}a>", f"<{EX}email>", '"a@x"')
    rows = db.query(
        f'SELECT ?n ?m WHERE {{ ?p <{EX}name> ?n . '
        f'OPTIONAL {{ ?p <{EX}email> ?m }} }}')
    assert sorted(rows) == [["Alice", "a@x"], ["Bob", ""]]


def test_optional_filter_inside_group():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}name>", '"Alice"')
    db.add_triple(f"<{EX}b>", f"<{EX}name>", '"Bob"')
    db.add_triple(f"<{EX}a>", f"<{EX}age>", '"17"')
    db.add_triple(f"<{EX}b>", f"<{EX}age>", '"42"')
    rows = db.query(
        f'SELECT ?n ?a WHERE {{ ?p <{EX}name> ?n . '
        f'OPTIONAL {{ ?p <{EX}age> ?a . FILTER(?a > 20) }} }}')
    assert sorted(rows) == [["Alice", ""], ["Bob", "42"]]


def test_optional_bound_filter_after():
    """!BOUND over an OPTIONAL var selects the unmatched rows."""
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}name>", '"Alice"')
    db.add_triple(f"<{EX}b>", f"<{EX}name>", '"Bob"')
    db.add_triple(f"<{EX}a>", f"<{EX}email>", '"a@x"')
    rows = db.query(
        f'SELECT ?n WHERE {{ ?p <{EX}name> ?n . '
        f'OPTIONAL {{ ?p <{EX}email> ?m }} FILTER(!BOUND(?m)) }}')
    assert rows == [["Bob"]]


def test_optional_multiple_matches_multiplicity():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}name>", '"Alice"')
    db.add_triple(f"<{EX}a>", f"<{EX}phone>", '"1"')
    db.add_triple(f"<{EX}a>", f"<{EX}phone>", '"2"')
    rows = db.query(
        f'SELECT ?n ?ph WHERE {{ ?p <{EX}name> ?n . '
        f'OPTIONAL {{ ?p <{EX}phone> ?ph }} }}')
    assert sorted(rows) == [["Alice", "1"], ["Alice", "2"]]


def test_ask_query():
    """ASK { ... } boolean queries (engine extension)."""
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", f"<{EX}b>")
    assert db.query(f'ASK {{ <{EX}a> <{EX}p> ?x }}') == [["true"]]
    assert db.query(f'ASK {{ <{EX}a> <{EX}q> ?x }}') == [["false"]]
    assert db.query(f'ASK {{ ?s <{EX}p> ?o . FILTER(?s = ?o) }}') == [["false"]]
    # cached second run
    assert db.query(f'ASK {{ <{EX}a> <{EX}p> ?x }}') == [["true"]]


def test_graph_variable_flows_into_update_templates():
    """ref sparql_graph_test.rs graph_variables_flow_from_where_into_
    delete_and_insert_templates: GRAPH ?g templates substitute per
    solution."""
    db = SparqlDatabase()
    db.query('INSERT DATA { GRAPH <http://g1> { <http://e/a> <http://e/p> <http://e/b> } }')
    db.query('INSERT DATA { GRAPH <http://g2> { <http://e/c> <http://e/p> <http://e/d> } }')
    db.query('''DELETE { GRAPH ?g { ?s <http://e/p> ?o } }
                INSERT { GRAPH ?g { ?s <http://e/q> ?o } }
                WHERE { GRAPH ?g { ?s <http://e/p> ?o } }''')
    assert sorted(db.query(
        'SELECT ?g ?s WHERE { GRAPH ?g { ?s <http://e/q> ?o } }')) == [
        ["http://g1", "http://e/a"], ["http://g2", "http://e/c"]]
    assert db.query(
        'SELECT ?s WHERE { GRAPH <http://g1> { ?s <http://e/p> ?o } }') == []


def test_construct_query():
    """Standalone CONSTRUCT (engine extension beyond the reference, which
    only uses CONSTRUCT inside RULE bodies)."""
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", f"<{EX}b>")
    db.add_triple(f"<{EX}b>", f"<{EX}p>", f"<{EX}c>")
    rows = db.query(
        f'CONSTRUCT {{ ?o <{EX}invP> ?s }} WHERE {{ ?s <{EX}p> ?o }}')
    assert sorted(rows) == [
        [f"{EX}b", f"{EX}invP", f"{EX}a"],
        [f"{EX}c", f"{EX}invP", f"{EX}b"]]
    one = db.query(
        f'CONSTRUCT {{ ?o <{EX}invP> ?s }} WHERE {{ ?s <{EX}p> ?o }} LIMIT 1')
    assert len(one) == 1


def test_describe_query():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", f"<{EX}b>")
    db.add_triple(f"<{EX}b>", f"<{EX}p>", f"<{EX}c>")
    rows = db.query(f'DESCRIBE <{EX}b>')
    assert sorted(rows) == [
        [f"{EX}a", f"{EX}p", f"{EX}b"],
        [f"{EX}b", f"{EX}p", f"{EX}c"]]
    assert db.query(f'DESCRIBE <{EX}missing>') == []


def test_property_path_sequence_and_inverse():
    """Path subset (engine extension): `p1/p2` sequences and `^p` inverse
    steps desugar to join chains."""
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}worksFor>", f"<{EX}d1>")
    db.add_triple(f"<{EX}d1>", f"<{EX}locatedIn>", f"<{EX}c1>")
    db.add_triple(f"<{EX}b>", f"<{EX}worksFor>", f"<{EX}d1>")
    rows = db.query(
        f'SELECT ?c WHERE {{ <{EX}a> <{EX}worksFor>/<{EX}locatedIn> ?c }}')
    assert rows == [[f"{EX}c1"]]
    rows = db.query(
        f'SELECT ?e WHERE {{ ?e <{EX}worksFor>/<{EX}locatedIn> <{EX}c1> }}')
    assert sorted(r[0] for r in rows) == [f"{EX}a", f"{EX}b"]
    rows = db.query(f'SELECT ?d WHERE {{ ?d ^<{EX}worksFor> <{EX}a> }}')
    assert rows == [[f"{EX}d1"]]
    # 3-step with inverse in the middle: a worksFor d1 ^worksFor b
    rows = db.query(
        f'SELECT ?x WHERE {{ <{EX}a> <{EX}worksFor>/^<{EX}worksFor> ?x }}')
    assert sorted(r[0] for r in rows) == [f"{EX}a", f"{EX}b"]


def test_property_path_transitive_closure():
    """p+ / p* closure paths (engine extension): materialized by the
    log-doubling device join, cached per store version, cycle-safe."""
    db = SparqlDatabase()
    for i in range(6):
        db.add_triple(f"<{EX}n{i}>", f"<{EX}next>", f"<{EX}n{i+1}>")
    rows = db.query(f'SELECT ?o WHERE {{ <{EX}n0> <{EX}next>+ ?o }}')
    assert sorted(r[0] for r in rows) == [f"{EX}n{i}" for i in range(1, 7)]
    assert len(db.query(f'SELECT ?s ?o WHERE {{ ?s <{EX}next>+ ?o }}')) == 21
    star = db.query(f'SELECT ?o WHERE {{ <{EX}n4> <{EX}next>* ?o }}')
    assert sorted(r[0] for r in star) == [f"{EX}n4", f"{EX}n5", f"{EX}n6"]
    # cycle: closure must terminate and be complete
    db.add_triple(f"<{EX}n6>", f"<{EX}next>", f"<{EX}n0>")
    assert len(db.query(f'SELECT ?s ?o WHERE {{ ?s <{EX}next>+ ?o }}')) == 49
    # closure composes with further joins
    db.add_triple(f"<{EX}n3>", f"<{EX}tag>", '"x"')
    rows = db.query(
        f'SELECT ?t WHERE {{ <{EX}n1> <{EX}next>+ ?m . ?m <{EX}tag> ?t }}')
    assert rows == [["x"]]


def test_having_clause():
    """HAVING over aggregates (engine extension)."""
    db = SparqlDatabase()
    for i in range(10):
        db.add_triple(f"<{EX}s{i}>", f"<{EX}grp>", f"<{EX}g{i % 3}>")
    rows = db.query(
        f'SELECT ?g (COUNT(*) AS ?c) WHERE {{ ?s <{EX}grp> ?g }} '
        f'GROUP BY ?g HAVING(?c > 3) ORDER BY ?g')
    assert rows == [[f"{EX}g0", "4"]]
    # single-group HAVING gates the whole result
    assert db.query(
        f'SELECT (COUNT(*) AS ?c) WHERE {{ ?s <{EX}grp> ?g }} '
        f'HAVING(?c > 100)') == []
    assert db.query(
        f'SELECT (COUNT(*) AS ?c) WHERE {{ ?s <{EX}grp> ?g }} '
        f'HAVING(?c > 5)') == [["10"]]


def test_const_subject_star_fusion():
    """>=2 (const-s, const-p, ?o) patterns fuse into one region fetch
    (PConstStar); multiset + repeated-var + missing-pattern semantics are
    preserved."""
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}name>", '"Alice"')
    db.add_triple(f"<{EX}a>", f"<{EX}sal>", '"5"')
    db.add_triple(f"<{EX}a>", f"<{EX}pos>", '"dev"')
    db.add_triple(f"<{EX}a>", f"<{EX}pos>", '"mgr"')
    db.add_triple(f"<{EX}b>", f"<{EX}name>", '"Bob"')
    rows = db.query(
        f'SELECT ?n ?s ?p WHERE {{ <{EX}a> <{EX}name> ?n ; '
        f'<{EX}sal> ?s ; <{EX}pos> ?p }}')
    assert sorted(rows) == [["Alice", "5", "dev"], ["Alice", "5", "mgr"]]
    assert db.query(
        f'SELECT ?n WHERE {{ <{EX}a> <{EX}name> ?n ; <{EX}nope> ?x }}') == []
    # repeated var constrains equality across the star
    assert db.query(
        f'SELECT ?x WHERE {{ <{EX}a> <{EX}name> ?x ; <{EX}sal> ?x }}') == []
    db.add_triple(f"<{EX}a>", f"<{EX}alias>", '"Alice"')
    assert db.query(
        f'SELECT ?x WHERE {{ <{EX}a> <{EX}name> ?x ; <{EX}alias> ?x }}') \
        == [["Alice"]]
    # plan shape: the fused op is actually used
    from kolibrie_amd.parsing.sparql import parse_combined_query
    from kolibrie_amd.plan.lower import build_logical_plan
    from kolibrie_amd.plan.optimizer import Streamertail
    from kolibrie_amd.plan.physical import PConstStar
    cq = parse_combined_query(
        f'SELECT ?n ?s WHERE {{ <{EX}a> <{EX}name> ?n ; <{EX}sal> ?s }}')
    plan = Streamertail(db.get_or_build_stats()).find_best_plan(
        build_logical_plan(cq.select.where, db, {}))
    found = []
    def walk(x):
        found.append(isinstance(x, PConstStar))
        for attr in ("left", "right", "input"):
            if hasattr(x, attr):
                walk(getattr(x, attr))
    walk(plan)
    assert any(found)


def test_group_count_pushdown_matches_generic():
    """Sorted-region GROUP BY pushdown (cached path) vs the generic
    aggregate path (uncached execute_select)."""
    from kolibrie_amd.engine.query import execute_select
    from kolibrie_amd.parsing.sparql import parse_combined_query
    db = SparqlDatabase()
    for i in range(40):
        db.add_triple(f"<{EX}s{i}>", f"<{EX}grp>", f"<{EX}g{i % 7}>")
        db.add_triple(f"<{EX}s{i}>", f"<{EX}other>", '"x"')
    q = (f'SELECT ?g (COUNT(*) AS ?c) WHERE {{ ?s <{EX}grp> ?g }} '
         f'GROUP BY ?g ORDER BY ?g')
    cached = db.query(q)
    cq = parse_combined_query(q)
    generic = execute_select(cq.select, db, dict(db.prefixes))
    assert cached == generic
    assert len(cached) == 7 and cached[0][1] in ("5", "6")
    # group by SUBJECT side too
    q2 = (f'SELECT ?s (COUNT(*) AS ?c) WHERE {{ ?s <{EX}grp> ?o }} '
          f'GROUP BY ?s ORDER BY ?s')
    cq2 = parse_combined_query(q2)
    assert db.query(q2) == execute_select(cq2.select, db, dict(db.prefixes))


def test_property_path_alternatives():
    """(p1|p2) alternatives (engine extension): desugared to a
    VALUES-constrained variable predicate; composes with sequences."""
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p1>", f"<{EX}x>")
    db.add_triple(f"<{EX}a>", f"<{EX}p2>", f"<{EX}y>")
    db.add_triple(f"<{EX}a>", f"<{EX}p3>", f"<{EX}z>")
    db.add_triple(f"<{EX}x>", f"<{EX}q>", f"<{EX}deep>")
    rows = db.query(
        f'SELECT ?o WHERE {{ <{EX}a> (<{EX}p1>|<{EX}p2>) ?o }}')
    assert sorted(r[0] for r in rows) == [f"{EX}x", f"{EX}y"]
    rows = db.query(
        f'SELECT ?o WHERE {{ <{EX}a> (<{EX}p1>|<{EX}p2>)/<{EX}q> ?o }}')
    assert rows == [[f"{EX}deep"]]


def test_property_path_closure_in_named_graph():
    """p+ closure scoped to GRAPH <iri>: only that graph's edges close."""
    db = SparqlDatabase()
    db.query(f'INSERT DATA {{ GRAPH <{EX}g> {{ <{EX}a> <{EX}n> <{EX}b> . '
             f'<{EX}b> <{EX}n> <{EX}c> }} }}')
    db.add_triple(f"<{EX}c>", f"<{EX}n>", f"<{EX}d>")  # default graph only
    rows = db.query(
        f'SELECT ?o WHERE {{ GRAPH <{EX}g> {{ <{EX}a> <{EX}n>+ ?o }} }}')
    assert sorted(r[0] for r in rows) == [f"{EX}b", f"{EX}c"]


def test_group_pushdown_respects_from_dataset():
    """GROUP BY pushdown must not bypass FROM dataset views."""
    db = SparqlDatabase()
    db.query(f'INSERT DATA {{ GRAPH <{EX}gX> {{ <{EX}s1> <{EX}grp> <{EX}g1> . '
             f'<{EX}s2> <{EX}grp> <{EX}g1> }} }}')
    db.add_triple(f"<{EX}s3>", f"<{EX}grp>", f"<{EX}g2>")
    q = (f'SELECT ?g (COUNT(*) AS ?c) FROM <{EX}gX> WHERE '
         f'{{ ?s <{EX}grp> ?g }} GROUP BY ?g ORDER BY ?g')
    assert db.query(q) == [[f"{EX}g1", "2"]]
    assert db.query(q) == [[f"{EX}g1", "2"]]  # cached run identical
    # default-graph query unaffected
    q2 = (f'SELECT ?g (COUNT(*) AS ?c) WHERE {{ ?s <{EX}grp> ?g }} '
          f'GROUP BY ?g ORDER BY ?g')
    assert db.query(q2) == [[f"{EX}g2", "1"]]


def test_query_columns_matches_rows():
    from kolibrie_amd import SparqlDatabase
    EX = "http://example.org/"
    db = SparqlDatabase()
    for i in range(200):
        db.add_triple(f"<{EX}e{i}>", f"<{EX}p>", f'"{i % 7}"')
    q = f"SELECT ?s ?o WHERE {{ ?s <{EX}p> ?o }} ORDER BY ?s ?o"
    rows = db.query(q)
    cols = db.query_columns(q)
    assert list(cols.keys()) == ["s", "o"]
    assert cols["s"] == [r[0] for r in rows]
    assert cols["o"] == [r[1] for r in rows]
    import pytest
    with pytest.raises(ValueError):
        db.query_columns(f"INSERT DATA {{ <{EX}x> <{EX}p> \"v\" }}")


def test_tracer_cpu_fallback_and_overhead_off():
    """Tracer disabled: no records accumulate; enabled on CPU it
    wall-clock-times each operator."""
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.engine import tracer

    db = SparqlDatabase(device="cpu")
    db.add_triple("<http://e/a>", "<http://e/p>", '"1"')
    tracer.reset()
    db.query("SELECT ?s WHERE { ?s <http://e/p> ?o }")
    assert tracer.snapshot() == {}
    tracer.enable()
    try:
        db.query("SELECT ?s WHERE { ?s <http://e/p> ?o }")
        snap = tracer.snapshot()
    finally:
        tracer.disable()
        tracer.reset()
    assert snap and all(ms >= 0.0 for _, ms in snap.values())
