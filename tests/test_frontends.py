"""Frontends: HTTP server endpoints, CLI, QueryBuilder, QueryEngine
(mirrors kolibrie-http-server behavior and querybuilder_test.rs)."""
import json

import pytest

from kolibrie_amd import SparqlDatabase

EX = "http://example.org/"


@pytest.fixture
def client():
    httpx = pytest.importorskip("httpx")
    pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient
    from kolibrie_amd.frontends.http_server import create_app
    db = SparqlDatabase()
    db.add_triple(f"<{EX}alice>", f"<{EX}name>", '"Alice"')
    db.add_triple(f"<{EX}bob>", f"<{EX}name>", '"Bob"')
    app = create_app(db)
    return TestClient(app)


def test_http_query(client):
    r = client.post("/query", content=f"SELECT ?s ?n WHERE {{ ?s <{EX}name> ?n }}")
    assert r.status_code == 200
    data = r.json()
    assert data["head"]["vars"] == ["s", "n"]
    vals = {b["n"]["value"] for b in data["results"]["bindings"]}
    assert vals == {"Alice", "Bob"}
    assert data["results"]["bindings"][0]["s"]["type"] == "uri"


def test_http_query_rejects_update(client):
    r = client.post("/query", content=f'INSERT DATA {{ <{EX}x> <{EX}p> "v" }}')
    assert r.status_code == 400


def test_http_update_endpoint(client):
    r = client.post("/update", content=f'INSERT DATA {{ <{EX}x> <{EX}p> "v" }}')
    assert r.status_code == 200
    r = client.post("/query", content=f"SELECT ?o WHERE {{ <{EX}x> <{EX}p> ?o }}")
    assert r.json()["results"]["bindings"][0]["o"]["value"] == "v"


def test_http_playground(client):
    r = client.get("/")
    assert r.status_code == 200
    assert "kolibrie_amd" in r.text


def test_http_rsp_session(client):
    q = f"""PREFIX ex: <{EX}>
REGISTER RSTREAM <http://out> AS
SELECT ?s ?o
FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 10 STEP 10]
WHERE {{ WINDOW <http://w1> {{ ?s ex:temp ?o }} }}"""
    r = client.post("/rsp/register", content=json.dumps({"query": q}))
    assert r.status_code == 200
    sid = r.json()["session"]
    events = [{"stream": "http://s1", "s": f"<{EX}m1>", "p": f"<{EX}temp>",
               "o": f'"{t}"', "ts": t} for t in range(0, 12)]
    r = client.post("/rsp/push", content=json.dumps(
        {"session": sid, "events": events}))
    assert r.status_code == 200


def test_http_rsp_query_stateless(client):
    q = f"""PREFIX ex: <{EX}>
REGISTER RSTREAM <http://out> AS
SELECT ?s ?o
FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 5 STEP 5]
WHERE {{ WINDOW <http://w1> {{ ?s ex:temp ?o }} }}"""
    events = [{"stream": "http://s1", "s": f"<{EX}m>", "p": f"<{EX}temp>",
               "o": f'"{t}"', "ts": t} for t in range(0, 7)]
    r = client.post("/rsp-query", content=json.dumps(
        {"query": q, "events": events}))
    assert r.status_code == 200
    assert r.json()["results"]


def test_cli(tmp_path, capsys):
    from kolibrie_amd.frontends.cli import main
    nt = tmp_path / "data.nt"
    nt.write_text(f'<{EX}a> <{EX}p> "v1" .\n<{EX}b> <{EX}p> "v2" .\n')
    rc = main(["--file", str(nt), "--query",
               f"SELECT ?s ?o WHERE {{ ?s <{EX}p> ?o }} ORDER BY ?o"])
    assert rc == 0
    out = capsys.readouterr().out.strip().split("\n")
    assert out == [f"{EX}a\tv1", f"{EX}b\tv2"]


def test_query_builder():
    from kolibrie_amd.engine.query_builder import QueryBuilder
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}name>", '"Alice"')
    db.add_triple(f"<{EX}b>", f"<{EX}name>", '"Bob"')
    db.add_triple(f"<{EX}a>", f"<{EX}age>", '"30"')
    rows = (QueryBuilder(db)
            .with_predicate(f"{EX}name")
            .with_object_starting("A")
            .execute())
    assert rows == [(f"{EX}a", f"{EX}name", "Alice")]
    rows = (QueryBuilder(db).with_subject(f"{EX}a")
            .order_by(lambda t: t[1]).execute())
    assert len(rows) == 2
    # cross-DB join on subject
    db2 = SparqlDatabase()
    db2.add_triple(f"<{EX}a>", f"<{EX}dept>", '"eng"')
    joined = QueryBuilder(db).with_predicate(f"{EX}name").join(db2, on="s")
    assert len(joined) == 1


def test_query_builder_streaming():
    from kolibrie_amd.engine.query_builder import QueryBuilder
    db = SparqlDatabase()
    qb = (QueryBuilder(db)
          .with_predicate("temp")
          .window(4, 4)
          .as_stream())
    for ts in range(0, 9):
        qb.add_stream_triple((f"m{ts % 2}", "temp", str(ts)), ts)
    res = qb.get_stream_results()
    assert len(res) == 2
    assert all(t[1] == "temp" for win in res for t in win)


def test_query_engine_explain():
    from kolibrie_amd.engine.query_engine import QueryEngine
    qe = QueryEngine()
    qe.add_triple(f"<{EX}a>", f"<{EX}p>", '"v"')
    assert qe.query(f"SELECT ?s WHERE {{ ?s <{EX}p> ?o }}") == [[f"{EX}a"]]
    plan = qe.explain(f"SELECT ?s WHERE {{ ?s <{EX}p> ?o . ?s <{EX}q> ?x }}")
    assert "Scan" in plan


def test_http_query_form_encoding(client):
    """SPARQL-protocol urlencoded form body (ref
    http_sparql_query_encodings_use_the_unified_query_executor)."""
    from urllib.parse import quote
    q = f'SELECT ?n WHERE {{ <{EX}alice> <{EX}name> ?n }}'
    r = client.post("/query", content=f"query={quote(q)}",
                    headers={"content-type":
                             "application/x-www-form-urlencoded"})
    assert r.status_code == 200
    vals = [b["n"]["value"] for b in r.json()["results"]["bindings"]]
    assert vals == ["Alice"]


def test_query_builder_parity_surface():
    """PyO3-surface parity: projections, group_by_* dicts, count,
    asc/desc, streaming config introspection."""
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.engine.query_builder import QueryBuilder

    db = SparqlDatabase(device="cpu")
    for i in range(6):
        db.add_triple(f"<http://e/s{i % 3}>", f"<http://e/p{i % 2}>",
                      f'"{i}"')
    qb = QueryBuilder(db).with_predicate_starting("http://e/p")
    assert qb.count() == 6
    assert sorted(set(qb.get_subjects())) == [
        "http://e/s0", "http://e/s1", "http://e/s2"]
    assert set(QueryBuilder(db).distinct().get_predicates()) == {
        "http://e/p0", "http://e/p1"}
    g = QueryBuilder(db).group_by_subject()
    assert len(g) == 3 and sum(len(v) for v in g.values()) == 6
    gp = QueryBuilder(db).group_by_predicate()
    assert len(gp) == 2
    rows = QueryBuilder(db).with_subject("http://e/s0").desc().execute()
    assert rows == sorted(rows, reverse=True)
    # predicate_ending filter
    assert QueryBuilder(db).with_predicate_ending("p1").count() == 3

    # streaming config surface
    qs = (QueryBuilder(db).window(10, 5).with_periodic_report(4)
          .with_tick_strategy("TimeDriven").with_stream_operator("RSTREAM"))
    assert qs.get_window_config() == (10, 5)
    assert qs.get_report_strategies() == ["Periodic"]
    assert qs.get_periodic_periods() == [4]
    assert qs.get_stream_operator() == "RSTREAM"
    assert not qs.is_streaming()
    qs.as_stream()
    assert qs.is_streaming()
    qs.add_stream_triple(("<a>", "<b>", "<c>"), 1)
    qs.stop_stream()
    assert not qs.is_streaming()
    qs.clear_stream_results()
    assert qs.get_all_stream_results() == []


def test_database_query_no_arg_returns_builder():
    """Reference parity: db.query() with no argument is the fluent
    builder entry point; db.query(sparql) executes."""
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.engine.query_builder import QueryBuilder

    db = SparqlDatabase(device="cpu")
    db.add_triple("<http://e/a>", "<http://e/p>", "<http://e/b>")
    qb = db.query()
    assert isinstance(qb, QueryBuilder)
    assert qb.count() == 1
    rows = db.query("SELECT ?s WHERE { ?s <http://e/p> ?o }")
    assert rows == [["http://e/a"]]


def test_compat_knowledge_graph_rule_surface():
    """Reference PyO3 KG parity: hand-built Rule(premise, filters,
    conclusion) with Term.Variable/Constant and FilterCondition; the
    infer_* entry points return decoded new triples."""
    from kolibrie_amd.compat import (FilterCondition, PyKnowledgeGraph,
                                     Rule_from_parts, Term, TriplePattern)
    from kolibrie_amd.storage.terms import Constant

    kg = PyKnowledgeGraph()
    kg.add_abox_triple("<http://e/a>", "<http://e/salary>", "50")
    kg.add_abox_triple("<http://e/b>", "<http://e/salary>", "10")
    sal = Constant(kg.encode_term("<http://e/salary>"))
    rich = Constant(kg.encode_term("<http://e/rich>"))
    yes = Constant(kg.encode_term("yes"))
    r = Rule_from_parts(
        [TriplePattern(Term.Variable("x"), sal, Term.Variable("s"))],
        [FilterCondition("s", ">", "20")],
        [TriplePattern(Term.Variable("x"), rich, yes)])
    kg.add_rule(r)
    derived = kg.infer_new_facts_semi_naive()
    assert derived == [("<http://e/a>", "<http://e/rich>", "yes")]
    # second run: fixpoint reached, nothing new
    assert kg.infer_new_facts_semi_naive() == []
    assert kg.query_abox(p="<http://e/rich>") == [
        ("<http://e/a>", "<http://e/rich>", "yes")]
    # variable-vs-variable filter: != compares dictionary ids
    same = Constant(kg.encode_term("<http://e/self_paid>"))
    pay = Constant(kg.encode_term("<http://e/pays>"))
    kg.add_abox_triple("<http://e/a>", "<http://e/pays>", "<http://e/a>")
    kg.add_abox_triple("<http://e/a>", "<http://e/pays>", "<http://e/b>")
    r2 = Rule_from_parts(
        [TriplePattern(Term.Variable("x"), pay, Term.Variable("y"))],
        [FilterCondition("x", "=", "y")],
        [TriplePattern(Term.Variable("x"), same, yes)])
    kg.add_rule(r2)
    new2 = kg.infer_new_facts_semi_naive()
    assert ("<http://e/a>", "<http://e/self_paid>", "yes") in new2


def test_cli_end_to_end(tmp_path, capsys):
    """CLI: --file + --query loads (format-sniffed), executes, prints TSV
    and JSON (ref cli/src/main.rs:15-41)."""
    from kolibrie_amd.frontends.cli import main

    nt = tmp_path / "data.nt"
    nt.write_text('<http://e/a> <http://e/p> "1" .\n'
                  '<http://e/b> <http://e/p> "2" .\n')
    rc = main(["-f", str(nt), "-q",
               "SELECT ?s ?o WHERE { ?s <http://e/p> ?o } ORDER BY ?o"])
    out = capsys.readouterr().out.strip().splitlines()
    assert rc == 0
    assert out == ["http://e/a\t1", "http://e/b\t2"]

    qf = tmp_path / "q.rq"
    qf.write_text("SELECT (COUNT(*) AS ?c) WHERE { ?s ?p ?o }")
    rc = main(["-f", str(nt), "--query-file", str(qf), "--format", "json"])
    import json
    assert rc == 0
    assert json.loads(capsys.readouterr().out) == [["2"]]


def test_query_engine_explain_renders_plan():
    """explain() renders the optimized physical plan with scan/join
    nodes and estimated orders (ref query_engine.rs explain)."""
    from kolibrie_amd.engine.query_engine import QueryEngine

    qe = QueryEngine()
    for i in range(20):
        qe.add_triple(f"<http://e/s{i}>", "<http://e/p>", f'"{i}"')
        qe.add_triple(f"<http://e/s{i}>", "<http://e/q>", "<http://e/t>")
    text = qe.explain(
        "SELECT ?s ?o WHERE { ?s <http://e/p> ?o . ?s <http://e/q> ?t }")
    assert "Scan" in text or "scan" in text
    assert "Join" in text or "join" in text or "Star" in text
    # the engine answers the same query it explained
    rows = qe.query(
        "SELECT (COUNT(*) AS ?c) WHERE { ?s <http://e/p> ?o . "
        "?s <http://e/q> ?t }")
    assert rows == [["20"]]
