"""Result-set equality across join shapes and join-mode choices
(ref kolibrie/tests/join_ordering_shapes_test.rs: chain/star datasets built
in-test; every planner choice must return the same multiset)."""
import os

import pytest

from kolibrie_amd import SparqlDatabase
from kolibrie_amd.engine.query import execute_query

EX = "http://e/"


def _chain_db(n=200):
    db = SparqlDatabase()
    for i in range(n):
        db.add_triple(f"<{EX}n{i}>", f"<{EX}next>", f"<{EX}n{i+1}>")
        db.add_triple(f"<{EX}n{i}>", f"<{EX}tag>", f'"{i % 7}"')
    return db


def _star_db(n=150):
    db = SparqlDatabase()
    for i in range(n):
        db.add_triple(f"<{EX}e{i}>", f"<{EX}name>", f'"name{i}"')
        db.add_triple(f"<{EX}e{i}>", f"<{EX}dept>", f"<{EX}d{i % 5}>")
        db.add_triple(f"<{EX}e{i}>", f"<{EX}grade>", f'"{i % 4}"')
    for d in range(5):
        db.add_triple(f"<{EX}d{d}>", f"<{EX}city>", f"<{EX}c{d % 3}>")
    return db


CHAIN_Q = f"""SELECT ?a ?c WHERE {{
    ?a <{EX}next> ?b . ?b <{EX}next> ?c . ?a <{EX}tag> "3" }}"""
STAR_Q = f"""SELECT ?e ?n ?city WHERE {{
    ?e <{EX}name> ?n . ?e <{EX}dept> ?d . ?e <{EX}grade> "2" .
    ?d <{EX}city> ?city }}"""


def _rows(db, q):
    return sorted(map(tuple, execute_query(q, db)))


@pytest.mark.parametrize("mode", ["auto", "hash", "bind"])
def test_chain_shape_same_results_across_modes(mode, monkeypatch):
    monkeypatch.setenv("KOLIBRIE_JOIN_MODE", mode)
    db = _chain_db()
    got = _rows(db, CHAIN_Q)
    monkeypatch.setenv("KOLIBRIE_JOIN_MODE", "auto")
    want = _rows(_chain_db(), CHAIN_Q)
    assert got == want and len(want) > 0


@pytest.mark.parametrize("mode", ["auto", "hash", "bind"])
def test_star_shape_same_results_across_modes(mode, monkeypatch):
    monkeypatch.setenv("KOLIBRIE_JOIN_MODE", mode)
    db = _star_db()
    got = _rows(db, STAR_Q)
    monkeypatch.setenv("KOLIBRIE_JOIN_MODE", "auto")
    want = _rows(_star_db(), STAR_Q)
    assert got == want and len(want) > 0


def test_pattern_order_invariance():
    """Shuffling the BGP's written pattern order never changes results."""
    db = _star_db()
    base = _rows(db, STAR_Q)
    reordered = f"""SELECT ?e ?n ?city WHERE {{
        ?d <{EX}city> ?city . ?e <{EX}grade> "2" .
        ?e <{EX}dept> ?d . ?e <{EX}name> ?n }}"""
    assert _rows(db, reordered) == base


def test_cartesian_fragment_still_correct():
    """Disconnected pattern groups multiply (cross product)."""
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", f"<{EX}b>")
    db.add_triple(f"<{EX}x>", f"<{EX}q>", f"<{EX}y>")
    db.add_triple(f"<{EX}x2>", f"<{EX}q>", f"<{EX}y>")
    rows = execute_query(
        f'SELECT ?s ?t WHERE {{ ?s <{EX}p> ?o . ?t <{EX}q> ?u }}', db)
    assert len(rows) == 2
