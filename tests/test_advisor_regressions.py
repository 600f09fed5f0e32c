"""Regressions for the round-1 advisor findings:

1. MINUS with disjoint variable domains must remove nothing (SPARQL spec:
   removal requires a compatible right row sharing >=1 bound variable).
2. OPTIONAL with no shared variables must not duplicate left rows.
3. MINUS must remove a left row whose shared key is PARTIALLY unbound when
   a keyed right row agrees on the left row's bound shared vars.
4. Binary checkpoints must round-trip '\r', literal backslash-n and other
   control characters inside terms.
"""
import pytest

from kolibrie_amd import SparqlDatabase
from kolibrie_amd.storage import checkpoint as cp

EX = "http://example.org/"


def _db(triples):
    db = SparqlDatabase()
    for s, p, o in triples:
        db.add_triple(s, p, o)
    return db


def test_minus_disjoint_domains_removes_nothing():
    db = _db([
        (f"<{EX}s1>", f"<{EX}p>", '"a"'),
        (f"<{EX}s2>", f"<{EX}p>", '"b"'),
        (f"<{EX}x>", f"<{EX}q>", '"c"'),
    ])
    rows = db.query(
        f"SELECT ?s ?o WHERE {{ ?s <{EX}p> ?o "
        f"MINUS {{ ?x <{EX}q> ?y }} }}")
    assert sorted(r[0] for r in rows) == [f"{EX}s1", f"{EX}s2"]


def test_minus_shared_var_still_removes():
    db = _db([
        (f"<{EX}s1>", f"<{EX}p>", '"a"'),
        (f"<{EX}s2>", f"<{EX}p>", '"b"'),
        (f"<{EX}s1>", f"<{EX}q>", '"c"'),
    ])
    rows = db.query(
        f"SELECT ?s WHERE {{ ?s <{EX}p> ?o MINUS {{ ?s <{EX}q> ?y }} }}")
    assert [r[0] for r in rows] == [f"{EX}s2"]


def test_optional_no_shared_vars_no_duplicates():
    db = _db([
        (f"<{EX}s1>", f"<{EX}p>", '"a"'),
        (f"<{EX}s2>", f"<{EX}p>", '"b"'),
        (f"<{EX}x>", f"<{EX}q>", '"c"'),
    ])
    rows = db.query(
        f"SELECT ?s ?y WHERE {{ ?s <{EX}p> ?o "
        f"OPTIONAL {{ ?x <{EX}q> ?y }} }}")
    # cartesian: each left row appears exactly once (1 right row), extended
    assert sorted((r[0], r[1]) for r in rows) == [
        (f"{EX}s1", "c"), (f"{EX}s2", "c")]


def test_optional_unmatched_rows_not_duplicated():
    db = _db([
        (f"<{EX}s1>", f"<{EX}p>", '"a"'),
        (f"<{EX}s2>", f"<{EX}p>", '"b"'),
        (f"<{EX}s1>", f"<{EX}q>", '"c"'),
    ])
    rows = db.query(
        f"SELECT ?s ?y WHERE {{ ?s <{EX}p> ?o "
        f"OPTIONAL {{ ?s <{EX}q> ?y }} }}")
    got = sorted((r[0], r[1]) for r in rows)
    assert got == [(f"{EX}s1", "c"), (f"{EX}s2", "")]


def test_minus_partially_unbound_key_removed():
    # OPTIONAL binds ?m only for s1; MINUS {?s <r> ?m} has shared vars
    # {?s, ?m}. Row (s2, UNBOUND) must still be removed when a right row
    # (s2, anything-agreeing-on-bound-vars) exists: dom intersection is
    # {?s}, non-empty.
    db = _db([
        (f"<{EX}s1>", f"<{EX}p>", '"a"'),
        (f"<{EX}s2>", f"<{EX}p>", '"b"'),
        (f"<{EX}s1>", f"<{EX}q>", '"m1"'),
        (f"<{EX}s2>", f"<{EX}r>", '"zz"'),
    ])
    rows = db.query(
        f"SELECT ?s WHERE {{ ?s <{EX}p> ?o "
        f"OPTIONAL {{ ?s <{EX}q> ?m }} "
        f"MINUS {{ ?s <{EX}r> ?m }} }}")
    # s1 has ?m="m1"; right rows for s1 under <r>: none -> s1 kept.
    # s2 has ?m unbound; right row (s2,"zz") agrees on ?s (the only bound
    # shared var) -> s2 removed.
    assert [r[0] for r in rows] == [f"{EX}s1"]


def test_minus_fully_unbound_shared_vars_kept():
    # left row whose ONLY shared var is unbound must be kept even when
    # right rows exist (empty dom intersection).
    db = _db([
        (f"<{EX}s1>", f"<{EX}p>", '"a"'),
        (f"<{EX}s2>", f"<{EX}p>", '"b"'),
        (f"<{EX}s1>", f"<{EX}q>", '"m1"'),
        (f"<{EX}anyone>", f"<{EX}r>", '"m1"'),
    ])
    rows = db.query(
        f"SELECT ?s WHERE {{ ?s <{EX}p> ?o "
        f"OPTIONAL {{ ?s <{EX}q> ?m }} "
        f"MINUS {{ ?z <{EX}r> ?m }} }}")
    # s1: ?m="m1" bound, right row binds ?m="m1" -> removed.
    # s2: ?m unbound -> dom intersection with right rows empty -> kept.
    assert [r[0] for r in rows] == [f"{EX}s2"]


def test_binary_checkpoint_control_chars(tmp_path):
    db = SparqlDatabase()
    nasty = ['"line1\rline2"', '"tab\there"', r'"back\nslash-n"',
             '"real\nnewline"', '"trailing\\\\"']
    for i, o in enumerate(nasty):
        db.add_triple(f"<{EX}s{i}>", f"<{EX}p>", o)
    path = str(tmp_path / "shard0.npz")
    cp.save_binary(db, path, rank=0)
    db2 = SparqlDatabase()
    cp.load_binary(db2, path)
    assert db2.dictionary.id_to_str == db.dictionary.id_to_str
    for i, o in enumerate(nasty):
        rows = db2.query(f"SELECT ?o WHERE {{ <{EX}s{i}> <{EX}p> ?o }}")
        assert len(rows) == 1


def test_binary_checkpoint_refuses_mismatched_dictionary(tmp_path):
    db = _db([(f"<{EX}a>", f"<{EX}p>", '"v"')])
    path = str(tmp_path / "shard0.npz")
    cp.save_binary(db, path, rank=0)
    db2 = SparqlDatabase()
    db2.add_triple(f"<{EX}other>", f"<{EX}term>", '"w"')
    with pytest.raises(ValueError):
        cp.load_binary(db2, path)
