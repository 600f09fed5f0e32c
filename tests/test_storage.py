"""Storage layer: dictionary, quoted triples, quad store, serializers.

Mirrors reference unit tests in shared/src/dataset_index.rs:593-825 and
dictionary/quoted-store behavior.
"""
import pytest
import torch

from kolibrie_amd.storage.dictionary import Dictionary, QuotedTripleStore
from kolibrie_amd.storage.dataset import DEFAULT_GRAPH, GraphIndex, QuadStore
from kolibrie_amd import SparqlDatabase


def test_dictionary_roundtrip():
    d = Dictionary()
    a = d.encode("http://example.org/a")
    b = d.encode("hello world")
    assert d.encode("http://example.org/a") == a
    assert d.decode(a) == "http://example.org/a"
    assert d.decode(b) == "hello world"
    assert d.lookup("missing") is None


def test_dictionary_numeric_values():
    d = Dictionary()
    x = d.encode("42.5")
    y = d.encode("not a number")
    assert d.numeric_value(x) == 42.5
    assert d.numeric_value(y) == 0.0


def test_dictionary_merge():
    d1 = Dictionary()
    d2 = Dictionary()
    a2 = d2.encode("alpha")
    b2 = d2.encode("beta")
    d1.encode("beta")
    remap = d1.merge(d2)
    assert d1.decode(remap[a2]) == "alpha"
    assert d1.decode(remap[b2]) == "beta"


def test_quoted_triple_store():
    q = QuotedTripleStore()
    qid = q.encode(1, 2, 3)
    assert qid & 0x8000_0000
    assert q.encode(1, 2, 3) == qid
    assert q.decode(qid) == (1, 2, 3)
    nested = q.encode(qid, 5, 6)
    assert q.decode(nested) == (qid, 5, 6)


def test_graph_index_lookup():
    s = [1, 1, 2, 3, 3, 3]
    p = [10, 11, 10, 10, 11, 11]
    o = [100, 101, 100, 102, 103, 104]
    gi = GraphIndex.from_columns(s, p, o, device="cpu")
    assert gi.n == 6
    ss, pp, oo = gi.lookup(1, None, None)
    assert ss.numel() == 2
    ss, pp, oo = gi.lookup(None, 11, None)
    assert sorted(oo.tolist()) == [101, 103, 104]
    ss, pp, oo = gi.lookup(None, None, 100)
    assert sorted(ss.tolist()) == [1, 2]
    ss, pp, oo = gi.lookup(3, 11, None)
    assert sorted(oo.tolist()) == [103, 104]
    assert gi.contains(3, 11, 104)
    assert not gi.contains(3, 11, 999)


def test_graph_index_dedup():
    gi = GraphIndex.from_columns([1, 1, 1], [2, 2, 2], [3, 3, 4], device="cpu")
    assert gi.n == 2


def test_quad_store_graphs():
    qs = QuadStore()
    qs.insert_quad(DEFAULT_GRAPH, 1, 2, 3)
    qs.insert_quad(7, 1, 2, 4)
    qs.insert_quad(7, 1, 2, 4)  # duplicate
    assert qs.triple_count() == 2
    assert qs.named_graph_ids() == [7]
    s, p, o = qs.query_graph(7, 1, None, None)
    assert o.tolist() == [4]
    qs.delete_quad(7, 1, 2, 4)
    assert qs.triple_count() == 1
    assert qs.named_graph_ids() == []


def test_quad_store_merged_dedup():
    qs = QuadStore()
    qs.insert_quad(1, 5, 6, 7)
    qs.insert_quad(2, 5, 6, 7)  # same triple, different graph
    qs.insert_quad(2, 5, 6, 8)
    merged = qs.merged_index([1, 2])
    assert merged.n == 2  # FROM-merge dedups


def test_graph_management():
    qs = QuadStore()
    qs.create_graph(9)
    assert 9 in qs.named_graph_ids()
    qs.insert_quad(9, 1, 1, 1)
    qs.clear_graph(9)
    assert qs.triple_count() == 0
    assert 9 in qs.named_graph_ids()  # catalog survives CLEAR
    assert qs.drop_graph(9)
    assert 9 not in qs.named_graph_ids()


def test_db_add_and_decode():
    db = SparqlDatabase()
    db.add_triple("<http://ex.org/alice>", "<http://ex.org/knows>", "<http://ex.org/bob>")
    db.add_triple("<http://ex.org/alice>", "<http://ex.org/name>", '"Alice"')
    trips = db.triples_as_strings()
    assert ("http://ex.org/alice", "http://ex.org/name", "Alice") in trips


def test_db_rdf_star_encode():
    db = SparqlDatabase()
    db.add_triple("<< <http://e/s> <http://e/p> <http://e/o> >>",
                  "<http://e/certainty>", '"0.9"')
    trips = db.triples_as_strings()
    assert len(trips) == 1
    assert trips[0][0].startswith("<<")
    assert "certainty" in trips[0][1]


def test_nquads_roundtrip():
    db = SparqlDatabase()
    db.add_quad_parts("<http://e/s>", "<http://e/p>", '"val"', "<http://e/g>")
    db.add_triple("<http://e/a>", "<http://e/b>", "<http://e/c>")
    text = db.generate_nquads()
    db2 = SparqlDatabase()
    db2.parse_nquads(text)
    assert db2.generate_nquads() == text


def test_union_reencoding():
    db1 = SparqlDatabase()
    db1.add_triple("<http://e/a>", "<http://e/p>", '"1"')
    db2 = SparqlDatabase()
    db2.add_triple("<http://e/b>", "<http://e/p>", '"2"')
    u = db1.union(db2)
    assert u.triple_count() == 2


def test_native_ntriples_bulk_parse_matches_python():
    """The C++ bulk N-Triples parser must agree with the Python tokenizer
    (differential)."""
    from kolibrie_amd.ops import _native
    if _native is None:
        import pytest
        pytest.skip("native extension not built")
    lines = []
    for i in range(500):
        lines.append(f'<http://e/s{i}> <http://e/p{i % 7}> "value {i} with \\"quote\\"" .')
        lines.append(f'<http://e/s{i}> <http://e/q> _:b{i} .')
    lines.append('<< <http://e/a> <http://e/b> <http://e/c> >> <http://e/cert> "0.9" .')
    lines.append('<http://e/lang> <http://e/label> "hello"@en .')
    lines.append('<http://e/typed> <http://e/num> "42"^^<http://www.w3.org/2001/XMLSchema#int> .')
    text = "\n".join(lines)
    db_native = SparqlDatabase()
    db_native.parse_ntriples(text)           # bulk path (text > 4096 bytes)
    db_py = SparqlDatabase()
    from kolibrie_amd.parsing.rdf_formats import _parse_ntriples_lines
    _parse_ntriples_lines(db_py, text.split("\n"))
    assert sorted(db_native.triples_as_strings()) == \
        sorted(db_py.triples_as_strings())
    assert db_native.triple_count() == db_py.triple_count()


def test_native_nquads_bulk_parse_matches_python():
    from kolibrie_amd.ops import _native
    if _native is None:
        pytest.skip("native extension not built")
    lines = []
    for i in range(400):
        g = f"<http://e/g{i % 5}>" if i % 3 else ""
        lines.append(f'<http://e/s{i}> <http://e/p{i % 7}> "v {i}" {g} .'.strip())
        lines.append(f'<http://e/s{i}> <http://e/q> _:b{i} <http://e/g1> .')
    lines.append('<< <http://e/a> <http://e/b> <http://e/c> >> <http://e/cert> "0.9" <http://e/g2> .')
    text = "\n".join(lines)
    db_native = SparqlDatabase()
    db_native.parse_nquads(text)               # bulk path (> 4096 bytes)
    db_py = SparqlDatabase()
    from kolibrie_amd.parsing.rdf_formats import _parse_nquads_lines
    _parse_nquads_lines(db_py, text.split("\n"))
    assert db_native.generate_nquads().count("\n") == \
        db_py.generate_nquads().count("\n")
    assert sorted(db_native.generate_nquads().split("\n")) == \
        sorted(db_py.generate_nquads().split("\n"))


def test_reference_api_surface():
    """Reference README core-method parity (README.md:737-888)."""
    db = SparqlDatabase()
    db.add_triple_parts("<http://e/s>", "<http://e/p>", "<http://e/o>")
    assert db.triple_count() == 1
    assert db.delete_triple_parts("<http://e/s>", "<http://e/p>", "<http://e/o>")
    assert db.triple_count() == 0
    db.add_triple("<http://e/a>", "<http://e/p>", '"x"')
    db.build_all_indexes()
    st = db.get_or_build_stats()
    db.invalidate_stats_cache()
    assert db.get_or_build_stats() is not st
    qb = db.query_builder()
    assert hasattr(qb, "with_subject")
    s = db.dictionary.lookup("http://e/a")
    p = db.dictionary.lookup("http://e/p")
    o = db.dictionary.lookup("x")
    assert db.decode_triple((s, p, o)) == ("http://e/a", "http://e/p", "x")
    assert db.decode_triple((s, p, 999999)) is None


def test_rdf_xml_roundtrip():
    """generate_rdf_xml output re-parses to the same triples."""
    db = SparqlDatabase()
    db.add_triple("<http://e/alice>", "<http://e/knows>", "<http://e/bob>")
    db.add_triple("<http://e/alice>", "<http://e/name>", '"Alice"')
    xml = db.generate_rdf_xml()
    db2 = SparqlDatabase()
    db2.parse_rdf(xml)
    assert sorted(db2.triples_as_strings()) == sorted(db.triples_as_strings())


def test_parse_ntriples_file_roundtrip(tmp_path):
    """File ingest (native parallel parser + in-C dictionary intern) must
    equal the in-memory parse path, including literal escapes and the
    value column for FILTER numerics."""
    from kolibrie_amd import SparqlDatabase
    EX = "http://example.org/"
    lines = []
    for i in range(500):
        lines.append(f'<{EX}e{i}> <{EX}salary> "{1000 + i}" .')
        lines.append(f'<{EX}e{i}> <{EX}name> "n\\"q{i}\\\\x" .')
    text = "\n".join(lines) + "\n"
    p = tmp_path / "x.nt"
    p.write_text(text)
    db1 = SparqlDatabase()
    db1.parse_ntriples_file(str(p))
    db2 = SparqlDatabase()
    db2.parse_ntriples(text)
    assert db1.triple_count() == db2.triple_count() == 1000
    q = (f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e <{EX}salary> ?s . "
         f"FILTER(?s >= 1250) }}")
    assert db1.query(q) == db2.query(q) == [["250"]]
    q2 = f'SELECT ?n WHERE {{ <{EX}e7> <{EX}name> ?n }}'
    assert db1.query(q2) == db2.query(q2)


def test_vocab_annex_semantics(tmp_path):
    """Bulk-vocabulary annex: ids from the native tail must behave exactly
    like python-dict ids — encode/lookup/decode/len/value-column/
    checkpoint round trip, plus post-annex incremental encodes."""
    import numpy as np
    import pytest
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.ops import _native
    from kolibrie_amd.storage import checkpoint as cp
    if _native is None:
        pytest.skip("native extension required")
    EX = "http://example.org/"
    lines = []
    for i in range(300):
        lines.append(f'<{EX}s{i}> <{EX}num> "{i * 3}" .')
        lines.append(f'<{EX}s{i}> <{EX}lab> "name {i}" .')
    p = tmp_path / "a.nt"
    p.write_text("\n".join(lines) + "\n")
    db = SparqlDatabase()
    db.parse_ntriples_file(str(p))
    d = db.dictionary
    assert d.annex is not None
    assert len(d) > 600
    # lookup/contains/decode through the annex
    sid = d.lookup(f"{EX}s7")
    assert sid is not None and d.decode(sid) == f"{EX}s7"
    assert d.contains("21") and not d.contains("nope-not-here")
    # numeric value column covers annex ids (FILTER semantics)
    rows = db.query(f"SELECT (COUNT(*) AS ?c) WHERE {{ ?s <{EX}num> ?v . "
                    f"FILTER(?v >= 450) }}")
    assert rows == [[str(sum(1 for i in range(300) if i * 3 >= 450))]]
    # post-annex incremental encode allocates non-colliding ids
    nid = d.encode("brand-new-term")
    assert d.decode(nid) == "brand-new-term"
    assert d.lookup("brand-new-term") == nid
    # row decode materializes annex strings
    out = db.query(f"SELECT ?s ?v WHERE {{ ?s <{EX}num> ?v }} "
                   f"ORDER BY ?s LIMIT 3")
    assert out[0][0].startswith(EX)
    # values_array covers the whole id space
    assert len(d.values_array()) == len(d)
    # checkpoint round trip through iter_strings
    path = str(tmp_path / "annexed.npz")
    cp.save_binary(db, path, rank=0)
    db2 = SparqlDatabase()
    cp.load_binary(db2, path)
    assert db2.query(f"SELECT (COUNT(*) AS ?c) WHERE {{ ?s ?p ?o }}") == \
        db.query(f"SELECT (COUNT(*) AS ?c) WHERE {{ ?s ?p ?o }}")
    assert db2.query(f'SELECT ?s WHERE {{ ?s <{EX}lab> "name 5" }}') == \
        db.query(f'SELECT ?s WHERE {{ ?s <{EX}lab> "name 5" }}')


def test_parse_ntriples_file_multichunk_merge(tmp_path):
    """Force the MULTI-chunk parallel path (>1 MiB file): cross-chunk
    duplicate terms (incl. escaped literals materialized per chunk) must
    dedup to single ids, and ids must be deterministic across loads of
    the same file."""
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.ops import _native
    import pytest
    if _native is None:
        pytest.skip("native extension required")
    EX = "http://example.org/padpadpadpadpadpadpadpadpadpad/"
    lines = []
    for i in range(12_000):
        # shared terms recur across the whole file -> land in many chunks
        lines.append(f'<{EX}e{i}> <{EX}type> <{EX}Employee> .')
        lines.append(f'<{EX}e{i}> <{EX}note> "esc\\tx\\"q\\\\{i % 7}" .')
    text = "\n".join(lines) + "\n"
    p = tmp_path / "big.nt"
    p.write_text(text)
    assert p.stat().st_size > (1 << 20)  # multi-chunk threshold

    db1 = SparqlDatabase()
    db1.parse_ntriples_file(str(p))
    assert db1.triple_count() == 24_000
    # shared constants resolved to ONE id each
    assert db1.query(
        f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e <{EX}type> <{EX}Employee> }}"
    ) == [["12000"]]
    # escaped literal dedup: 7 distinct notes
    assert db1.query(
        f"SELECT (COUNT(DISTINCT ?n) AS ?c) WHERE {{ ?e <{EX}note> ?n }}"
    ) == [["7"]]
    # escape decoding round-trips
    out = db1.query(f'SELECT ?n WHERE {{ <{EX}e3> <{EX}note> ?n }}')
    assert out == [['esc\tx"q\\3']]

    # determinism: same file + shard count -> identical annex ids
    db2 = SparqlDatabase()
    db2.parse_ntriples_file(str(p))
    assert db2.dictionary.lookup(f"{EX}e77") == \
        db1.dictionary.lookup(f"{EX}e77")
    assert db2.dictionary.lookup('esc\tx"q\\5') == \
        db1.dictionary.lookup('esc\tx"q\\5')


def test_bulk_load_dedups_against_python_prefix(tmp_path):
    """Terms already interned in the Python dictionary (including "" at
    id 0) must resolve to their EXISTING ids during a native bulk load —
    a duplicate annex id would make constant-term queries miss the
    bulk-loaded rows.  (Found by property-based fuzzing: the empty
    literal crashed the bump arena and then exposed the prefix-dedup
    hole.)"""
    import pytest
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.ops import _native
    if _native is None:
        pytest.skip("native extension required")
    p = tmp_path / "d.nt"
    p.write_text('<a> <b> "" .\n<c> <b> "" .\n<a> <b> "x" .\n')
    db = SparqlDatabase()
    db.add_triple("<a>", "<pre>", '"seen-before"')  # pre-bulk interning
    pre_id = db.dictionary.lookup("a")
    db.parse_ntriples_file(str(p))
    assert db.triple_count() == 4
    # "" (Python prefix id 0) matches bulk-loaded rows
    assert db.query('SELECT ?s WHERE { ?s ?p "" } ORDER BY ?s') == \
        [["a"], ["c"]]
    # the pre-interned subject unifies with its bulk occurrences
    assert db.query('SELECT ?o WHERE { <a> <b> ?o } ORDER BY ?o') == \
        [[""], ["x"]]
    # no duplicate id was allocated
    assert db.dictionary.lookup("a") == pre_id
    mod, h = db.dictionary.annex
    assert mod.vocab_lookup(h, "a") == pre_id
