"""Checkpoint/resume round trips (persistence = serialization
round trips, legacy-tolerant loads)."""
import pytest

from kolibrie_amd import SparqlDatabase
from kolibrie_amd.storage import checkpoint as cp

EX = "http://example.org/"


def _db():
    db = SparqlDatabase()
    db.add_triple(f"<{EX}a>", f"<{EX}p>", '"v1"')
    db.add_quad_parts(f"<{EX}b>", f"<{EX}p>", '"v2"', f"<{EX}g>")
    db.add_triple(f"<< <{EX}s> <{EX}q> <{EX}o> >>", f"<{EX}cert>", '"0.9"')
    return db


def test_nquads_checkpoint_roundtrip(tmp_path):
    db = _db()
    path = str(tmp_path / "dump.nq")
    cp.save_nquads(db, path)
    db2 = SparqlDatabase()
    cp.load_nquads(db2, path)
    assert db2.generate_nquads() == db.generate_nquads()


def test_binary_checkpoint_roundtrip(tmp_path):
    db = _db()
    path = str(tmp_path / "shard0.npz")
    cp.save_binary(db, path, rank=0)
    db2 = SparqlDatabase()
    meta = cp.load_binary(db2, path)
    assert meta["rank"] == 0
    assert db2.triple_count() == db.triple_count()
    rows = db2.query(f"SELECT ?o WHERE {{ <{EX}a> <{EX}p> ?o }}")
    assert rows == [["v1"]]
    rows = db2.query(
        f"SELECT ?c WHERE {{ << <{EX}s> <{EX}q> <{EX}o> >> <{EX}cert> ?c }}")
    assert rows == [["0.9"]]
