"""Filter-scope and subquery-modifier semantics (mirrors
sparql_filter_scope_test.rs and sparql_subquery_modifiers_test.rs)."""
import pytest

from kolibrie_amd import SparqlDatabase

EX = "http://example.org/"


def _db():
    db = SparqlDatabase()
    for name, age in (("alice", 30), ("bob", 45), ("carol", 22), ("dan", 45)):
        db.add_triple(f"<{EX}{name}>", f"<{EX}age>", f'"{age}"')
        db.add_triple(f"<{EX}{name}>", f"<{EX}name>", f'"{name}"')
    return db


def test_filter_scopes_to_its_group():
    db = _db()
    # FILTER inside the union branch applies only to that branch
    rows = db.query(f"""SELECT ?n WHERE {{
        {{ ?x <{EX}age> ?a . FILTER(?a > 40) ?x <{EX}name> ?n }}
        UNION
        {{ ?x <{EX}age> "22" . ?x <{EX}name> ?n }}
    }}""")
    assert sorted(r[0] for r in rows) == ["bob", "carol", "dan"]


def test_filter_sees_whole_group_even_when_written_first():
    db = _db()
    rows = db.query(f"""SELECT ?n WHERE {{
        FILTER(?a > 40)
        ?x <{EX}age> ?a . ?x <{EX}name> ?n
    }}""")
    assert sorted(r[0] for r in rows) == ["bob", "dan"]


def test_filter_on_outer_does_not_see_subquery_internals():
    db = _db()
    # subquery projects only ?x: outer filter on ?a has no binding -> empty
    rows = db.query(f"""SELECT ?x WHERE {{
        {{ SELECT ?x WHERE {{ ?x <{EX}age> ?a }} }}
        FILTER(?a > 0)
    }}""")
    assert rows == []


def test_subquery_limit_applies_before_outer_join():
    db = _db()
    rows = db.query(f"""SELECT ?n WHERE {{
        {{ SELECT ?x WHERE {{ ?x <{EX}age> ?a }} ORDER BY ?a LIMIT 1 }}
        ?x <{EX}name> ?n
    }}""")
    assert rows == [["carol"]]


def test_subquery_distinct_and_aggregate():
    db = _db()
    rows = db.query(f"""SELECT ?c WHERE {{
        {{ SELECT (COUNT(?x) AS ?c) WHERE {{ ?x <{EX}age> "45" }} }}
    }}""")
    assert rows == [["2"]]
    rows = db.query(f"""SELECT ?a WHERE {{
        {{ SELECT DISTINCT ?a WHERE {{ ?x <{EX}age> ?a }} }}
    }} ORDER BY ?a""")
    assert [r[0] for r in rows] == ["22", "30", "45"]


def test_order_limit_offset_pipeline():
    db = _db()
    rows = db.query(f"""SELECT ?n WHERE {{
        ?x <{EX}name> ?n . ?x <{EX}age> ?a
    }} ORDER BY DESC(?a) ?n LIMIT 2 OFFSET 1""")
    assert [r[0] for r in rows] == ["dan", "alice"]


def test_graph_visibility_rules():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}d>", f"<{EX}p>", '"default"')
    db.add_quad_parts(f"<{EX}n>", f"<{EX}p>", '"named"', f"<{EX}g>")
    # plain BGP sees only the default graph
    rows = db.query(f"SELECT ?o WHERE {{ ?s <{EX}p> ?o }}")
    assert [r[0] for r in rows] == ["default"]
    # GRAPH <g> sees only that graph
    rows = db.query(f"SELECT ?o WHERE {{ GRAPH <{EX}g> {{ ?s <{EX}p> ?o }} }}")
    assert [r[0] for r in rows] == ["named"]
    # FROM NAMED restricts GRAPH ?g iteration
    db.add_quad_parts(f"<{EX}n2>", f"<{EX}p>", '"other"', f"<{EX}g2>")
    rows = db.query(f"""SELECT ?o FROM <{EX}g> FROM NAMED <{EX}g2>
        WHERE {{ GRAPH ?g {{ ?s <{EX}p> ?o }} }}""")
    assert [r[0] for r in rows] == ["other"]
