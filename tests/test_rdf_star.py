"""RDF-star parity suite (mirrors kolibrie/tests/rdf_star_test.rs: quoted
triples in data and patterns, annotation syntax, TRIPLE/SUBJECT/PREDICATE/
OBJECT/isTRIPLE builtins, nesting, updates)."""
import pytest

from kolibrie_amd import SparqlDatabase
from kolibrie_amd.engine.query import execute_query

EX = "http://example.org/"


def _db_with_annotation():
    db = SparqlDatabase()
    db.parse_turtle(f"""
@prefix ex: <{EX}> .
ex:alice ex:knows ex:bob {{| ex:certainty "0.9" ; ex:source ex:wiki |}} .
ex:bob ex:knows ex:carol .
""")
    return db


def test_annotation_asserts_base_triple():
    db = _db_with_annotation()
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?o WHERE {{ ex:alice ex:knows ?o }}', db)
    assert rows == [[EX + "bob"]]


def test_quoted_subject_pattern_match():
    db = _db_with_annotation()
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?c WHERE {{ '
        f'<< ex:alice ex:knows ex:bob >> ex:certainty ?c }}', db)
    assert rows == [["0.9"]]


def test_quoted_pattern_with_inner_variable():
    db = _db_with_annotation()
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?who ?c WHERE {{ '
        f'<< ex:alice ex:knows ?who >> ex:certainty ?c }}', db)
    assert rows == [[EX + "bob", "0.9"]]


def test_quoted_pattern_all_variables():
    db = _db_with_annotation()
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?s ?p ?o WHERE {{ '
        f'<< ?s ?p ?o >> ex:source ex:wiki }}', db)
    assert rows == [[EX + "alice", EX + "knows", EX + "bob"]]


def test_two_annotations_on_same_triple():
    db = _db_with_annotation()
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?c ?src WHERE {{ '
        f'<< ex:alice ex:knows ex:bob >> ex:certainty ?c . '
        f'<< ex:alice ex:knows ex:bob >> ex:source ?src }}', db)
    assert rows == [["0.9", EX + "wiki"]]


def test_nested_quoted_triple():
    db = SparqlDatabase()
    db.add_triple(f"<< << <{EX}s> <{EX}p> <{EX}o> >> <{EX}saidBy> <{EX}me> >>",
                  f"<{EX}certainty>", '"0.5"')
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?c WHERE {{ '
        f'<< << ex:s ex:p ex:o >> ex:saidBy ex:me >> ex:certainty ?c }}', db)
    assert rows == [["0.5"]]


def test_quoted_object_position():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}stmt1>", f"<{EX}states>",
                  f"<< <{EX}a> <{EX}b> <{EX}c> >>")
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?s WHERE {{ '
        f'?s ex:states << ex:a ex:b ex:c >> }}', db)
    assert rows == [[EX + "stmt1"]]


def test_istriple_filter():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}stmt1>", f"<{EX}v>", f"<< <{EX}a> <{EX}b> <{EX}c> >>")
    db.add_triple(f"<{EX}stmt2>", f"<{EX}v>", f"<{EX}plain>")
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?s WHERE {{ ?s ex:v ?o . '
        f'FILTER(isTRIPLE(?o)) }}', db)
    assert rows == [[EX + "stmt1"]]
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?s WHERE {{ ?s ex:v ?o . '
        f'FILTER(!isTRIPLE(?o)) }}', db)
    assert rows == [[EX + "stmt2"]]


def test_bind_triple_constructor_and_accessors():
    db = _db_with_annotation()
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?s WHERE {{ '
        f'?x ex:certainty ?c . BIND(SUBJECT(?x) AS ?s) }}', db)
    assert rows == [[EX + "alice"]]
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?p ?o WHERE {{ ?x ex:certainty ?c . '
        f'BIND(PREDICATE(?x) AS ?p) BIND(OBJECT(?x) AS ?o) }}', db)
    assert rows == [[EX + "knows", EX + "bob"]]


def test_bind_triple_builds_matching_id():
    db = _db_with_annotation()
    # TRIPLE(s,p,o) must intern to the SAME quoted id as the data path
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?c WHERE {{ '
        f'?s ex:knows ?o . BIND(TRIPLE(?s, ex:knows, ?o) AS ?t) . '
        f'?t ex:certainty ?c }}', db)
    assert rows == [["0.9"]]


def test_insert_quoted_triple_via_update():
    db = SparqlDatabase()
    execute_query(
        f'PREFIX ex: <{EX}> INSERT DATA {{ '
        f'<< ex:x ex:y ex:z >> ex:conf "0.7" }}', db)
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?c WHERE {{ '
        f'<< ex:x ex:y ex:z >> ex:conf ?c }}', db)
    assert rows == [["0.7"]]


def test_delete_annotation_keeps_base():
    db = _db_with_annotation()
    execute_query(
        f'PREFIX ex: <{EX}> DELETE DATA {{ '
        f'<< ex:alice ex:knows ex:bob >> ex:certainty "0.9" }}', db)
    assert execute_query(
        f'PREFIX ex: <{EX}> SELECT ?c WHERE {{ '
        f'<< ex:alice ex:knows ex:bob >> ex:certainty ?c }}', db) == []
    assert execute_query(
        f'PREFIX ex: <{EX}> SELECT ?o WHERE {{ ex:alice ex:knows ?o }}', db) \
        == [[EX + "bob"]]


def test_quoted_roundtrip_serialization():
    db = _db_with_annotation()
    text = db.generate_nquads()
    db2 = SparqlDatabase()
    db2.parse_nquads(text)
    assert sorted(db2.triples_as_strings()) == sorted(db.triples_as_strings())


def test_annotation_join_with_filter_on_value():
    db = SparqlDatabase()
    db.parse_turtle(f"""
@prefix ex: <{EX}> .
ex:a ex:knows ex:b {{| ex:certainty "0.9" |}} .
ex:b ex:knows ex:c {{| ex:certainty "0.2" |}} .
""")
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?s ?o WHERE {{ '
        f'<< ?s ex:knows ?o >> ex:certainty ?c . FILTER(?c > 0.5) }}', db)
    assert rows == [[EX + "a", EX + "b"]]


def test_quoted_triples_distinct_from_plain():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}b>", f"<{EX}c>")
    db.add_triple(f"<< <{EX}a> <{EX}b> <{EX}c> >>", f"<{EX}meta>", '"m"')
    # plain pattern must not match the quoted-subject row
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?s WHERE {{ ?s ex:meta ?m }}', db)
    assert rows == [[f"<<<{EX}a> <{EX}b> <{EX}c>>>"]]
    rows = execute_query(
        f'PREFIX ex: <{EX}> SELECT ?p WHERE {{ ex:a ?p ?o }}', db)
    assert rows == [[EX + "b"]]
