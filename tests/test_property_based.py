"""Property-based tests (hypothesis): randomized invariants over the
string/ingest/window layers — deeper than the fixed-example unit tests.

Bounded examples keep the CPU suite fast; every property mirrors an
invariant the engine's correctness depends on.
"""
import string

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402


# term text that survives N-Triples framing: IRIs exclude '>', literals
# get escaped by the writer below
_iri_chars = st.text(
    alphabet=string.ascii_letters + string.digits + "/#._-~:%",
    min_size=1, max_size=24)
_lit_chars = st.text(
    alphabet=string.printable.replace("\r", "").replace("\x0b", "")
    .replace("\x0c", ""), min_size=0, max_size=24)


def _esc(lit: str) -> str:
    return (lit.replace("\\", "\\\\").replace('"', '\\"')
            .replace("\n", "\\n").replace("\t", "\\t"))


@settings(max_examples=30, deadline=None)
@given(st.lists(st.tuples(_iri_chars, _iri_chars,
                          st.one_of(_iri_chars.map(lambda x: ("iri", x)),
                                    _lit_chars.map(lambda x: ("lit", x)))),
                min_size=1, max_size=40))
def test_nt_file_ingest_equals_text_parse(rows):
    """The native file parser (annex path) must agree with the in-memory
    Python parse on arbitrary escaped content: same triple count and the
    same decoded rows."""
    import os
    import tempfile
    from kolibrie_amd import SparqlDatabase

    lines = []
    for s, p, (kind, o) in rows:
        o_txt = f"<http://o/{o}>" if kind == "iri" else f'"{_esc(o)}"'
        lines.append(f"<http://s/{s}> <http://p/{p}> {o_txt} .")
    text = "\n".join(lines) + "\n"

    db_text = SparqlDatabase()
    db_text.parse_ntriples(text)
    db_file = SparqlDatabase()
    fd, path = tempfile.mkstemp(suffix=".nt")
    try:
        with os.fdopen(fd, "w", encoding="utf-8") as f:
            f.write(text)
        db_file.parse_ntriples_file(path)
    finally:
        os.unlink(path)

    assert db_file.triple_count() == db_text.triple_count()
    q = "SELECT ?s ?p ?o WHERE { ?s ?p ?o } ORDER BY ?s ?p ?o"
    assert db_file.query(q) == db_text.query(q)


@settings(max_examples=30, deadline=None)
@given(st.lists(st.lists(st.integers(min_value=0, max_value=60),
                         min_size=1, max_size=8),
                min_size=1, max_size=6),
       st.integers(min_value=1, max_value=12),
       st.integers(min_value=1, max_value=8))
def test_window_range_views_equal_mask_path(ts_batches, width, slide):
    """K7 zero-copy scoping must equal the mask fallback for ANY batch
    timestamps (sorted or not), any width/slide."""
    import torch
    from kolibrie_amd.rsp.ring import DeviceStreamWindow

    def run(force_mask):
        w = DeviceStreamWindow(width=width, slide=slide, device="cpu")
        fired = []
        w.register_callback(
            lambda c: fired.append(
                (c.open, c.close, sorted(c.ts.tolist()),
                 sorted(c.s.tolist()))))
        if force_mask:
            w._sorted = False
        n = 0
        for ts in ts_batches:
            k = len(ts)
            s = torch.arange(n, n + k, dtype=torch.int32)
            n += k
            w.add_batch(s, s.clone(), s.clone(),
                        torch.tensor(ts, dtype=torch.int64))
        return fired

    assert run(False) == run(True)


@settings(max_examples=20, deadline=None)
@given(st.lists(st.text(min_size=0, max_size=30), min_size=1, max_size=50))
def test_checkpoint_dictionary_roundtrip(strings):
    """Binary checkpoint must round-trip ANY interned strings (controls,
    newlines, backslashes, unicode) with ids preserved."""
    import tempfile
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.storage import checkpoint as cp

    db = SparqlDatabase()
    ids = [db.dictionary.encode(s) for s in strings]
    db.add_triple("<http://a>", "<http://b>", "<http://c>")
    with tempfile.TemporaryDirectory() as d:
        path = f"{d}/ck.npz"
        cp.save_binary(db, path)
        db2 = SparqlDatabase()
        cp.load_binary(db2, path)
    for s, i in zip(strings, ids):
        assert db2.dictionary.decode(i) == s
        assert db2.dictionary.lookup(s) == i


@settings(max_examples=30, deadline=None)
@given(st.lists(st.text(alphabet=string.printable, min_size=0, max_size=40),
                min_size=1, max_size=200))
def test_dictionary_encode_bijection(strings):
    """encode is injective over distinct strings and decode inverts it,
    including via encode_many."""
    from kolibrie_amd.storage.dictionary import Dictionary

    d = Dictionary()
    ids = d.encode_many(strings)
    for s, i in zip(strings, ids.tolist()):
        assert d.decode(i) == s
        assert d.encode(s) == i
    distinct = len(set(strings) | {""})
    assert len(d) == distinct


@settings(max_examples=30, deadline=None)
@given(st.lists(st.lists(st.tuples(st.integers(0, 20), st.integers(0, 20)),
                         min_size=0, max_size=25),
                min_size=1, max_size=6),
       st.sampled_from(["RSTREAM", "ISTREAM", "DSTREAM"]))
def test_r2s_columnar_equals_host_sets(frames, mode):
    """K10 columnar R2S diff must emit the same SETS as the host
    tuple-set operator for any frame sequence."""
    import torch
    from kolibrie_amd.rsp.r2s import Relation2StreamOperator

    host = Relation2StreamOperator(mode)
    dev = Relation2StreamOperator(mode)
    for frame in frames:
        got_host = set(map(tuple, host.eval([tuple(t) for t in frame])))
        if frame:
            s = torch.tensor([t[0] for t in frame], dtype=torch.int32)
            o = torch.tensor([t[1] for t in frame], dtype=torch.int32)
            cols = dev.eval_columns([s, o])
        else:
            cols = dev.eval_columns([
                torch.empty(0, dtype=torch.int32),
                torch.empty(0, dtype=torch.int32)])
        got_dev = (set(zip(cols[0].tolist(), cols[1].tolist()))
                   if cols else set())
        assert got_dev == got_host, (mode, frame)


@settings(max_examples=15, deadline=None)
@given(st.lists(st.tuples(st.integers(0, 6), st.integers(0, 6))
                .filter(lambda e: e[0] < e[1]),   # DAG edges: SLD depth
                min_size=1, max_size=12, unique=True))  # stays bounded
def test_backward_chaining_agrees_with_forward_closure(edges):
    """For the canonical transitive rule, backward goal resolution must
    return exactly the forward-materialized answers (SLD vs semi-naive
    oracle property)."""
    from kolibrie_amd.reasoning.reasoner import Reasoner
    from kolibrie_amd.reasoning.rule import Rule
    from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable

    r = Reasoner()
    for a, b in edges:
        r.add_abox_triple(f"n{a}", "edge", f"n{b}")
    pid = r._i32(r.dictionary.encode("path"))
    eid = r._i32(r.dictionary.encode("edge"))
    r.add_rule(Rule(
        premise=[TriplePattern(Variable("x"), Constant(eid), Variable("y"))],
        conclusion=[TriplePattern(Variable("x"), Constant(pid),
                                  Variable("y"))]))
    r.add_rule(Rule(
        premise=[TriplePattern(Variable("x"), Constant(pid), Variable("y")),
                 TriplePattern(Variable("y"), Constant(eid), Variable("z"))],
        conclusion=[TriplePattern(Variable("x"), Constant(pid),
                                  Variable("z"))]))

    # forward oracle: python transitive closure over the edge set
    import itertools
    reach = set(edges)
    changed = True
    while changed:
        changed = False
        for (a, b), (c, d) in itertools.product(list(reach), list(reach)):
            if b == c and (a, d) not in reach:
                reach.add((a, d))
                changed = True

    start = edges[0][0]
    want = sorted({d for (a, d) in reach if a == start})
    got = r.backward_chaining((f"n{start}", "path", "?z"), max_depth=16)
    got_ids = sorted({r.dictionary.decode(b["z"]) for b in got})
    assert got_ids == [f"n{d}" for d in want], (edges, start)


@settings(max_examples=25, deadline=None)
@given(st.lists(st.tuples(st.integers(0, 5),
                          st.integers(-1000, 1000)),
                min_size=1, max_size=60))
def test_group_aggregates_match_python_oracle(rows):
    """GROUP BY COUNT/SUM/MIN/MAX/AVG through the full engine must equal
    a direct Python computation over the same groups."""
    from collections import defaultdict
    from kolibrie_amd import SparqlDatabase

    db = SparqlDatabase()
    # dedup: the store is a SET of triples
    uniq = {(g, v) for g, v in rows}
    for g, v in uniq:
        db.add_triple(f"<http://e/x{g}_{v}>", "<http://e/in>",
                      f"<http://g/{g}>")
        db.add_triple(f"<http://e/x{g}_{v}>", "<http://e/val>", f'"{v}"')
    got = db.query(
        "SELECT ?g (COUNT(?v) AS ?c) (SUM(?v) AS ?s) (MIN(?v) AS ?mn) "
        "(MAX(?v) AS ?mx) WHERE { ?x <http://e/in> ?g . "
        "?x <http://e/val> ?v } GROUP BY ?g ORDER BY ?g")
    groups = defaultdict(list)
    for g, v in uniq:
        groups[f"http://g/{g}"].append(v)
    want = [[g, str(len(vs)), str(sum(vs)), str(min(vs)), str(max(vs))]
            for g, vs in sorted(groups.items())]
    def num(x):
        f = float(x)
        return int(f) if f == int(f) else f
    got_n = [[r[0]] + [num(c) for c in r[1:]] for r in got]
    want_n = [[r[0]] + [num(c) for c in r[1:]] for r in want]
    assert got_n == want_n


@settings(max_examples=25, deadline=None)
@given(st.lists(st.integers(-500, 500), min_size=1, max_size=40,
                unique=True),
       st.integers(0, 10), st.integers(0, 10), st.booleans())
def test_order_limit_offset_semantics(vals, limit, offset, desc):
    """ORDER BY (numeric literals) + LIMIT/OFFSET must equal Python
    sorted()[offset:offset+limit]; DISTINCT removes duplicates."""
    from kolibrie_amd import SparqlDatabase

    db = SparqlDatabase()
    for i, v in enumerate(vals):
        db.add_triple(f"<http://e/s{i}>", "<http://e/v>", f'"{v}"')
        # duplicate object rows for the DISTINCT check
        db.add_triple(f"<http://e/dup{i}>", "<http://e/v>", f'"{v}"')
    direction = "DESC(?v)" if desc else "?v"
    got = db.query(
        f"SELECT ?v WHERE {{ ?s <http://e/v> ?v }} ORDER BY {direction} "
        f"LIMIT {limit} OFFSET {offset}")
    ordered = sorted(vals * 2, reverse=desc)
    want = [[str(v)] for v in ordered[offset:offset + limit]]
    assert [[r[0]] for r in got] == want

    got_d = db.query(
        "SELECT DISTINCT ?v WHERE { ?s <http://e/v> ?v }")
    assert sorted(int(r[0]) for r in got_d) == sorted(vals)


# recursive RDF-star term strategy: IRIs at the leaves, quoted triples
# nesting up to depth 3
_star_leaf = st.text(alphabet=string.ascii_lowercase + string.digits,
                     min_size=1, max_size=8).map(lambda x: f"<http://t/{x}>")
_star_term = st.recursive(
    _star_leaf,
    lambda inner: st.tuples(inner, _star_leaf, inner).map(
        lambda t: f"<< {t[0]} {t[1]} {t[2]} >>"),
    max_leaves=6)


@settings(max_examples=30, deadline=None)
@given(_star_term)
def test_rdf_star_encode_decode_roundtrip(term):
    """encode_term_star -> decode_term -> encode_term_star must be the
    identity for arbitrarily nested quoted triples, and interning is
    idempotent (same id on re-encode of the original text)."""
    from kolibrie_amd import SparqlDatabase

    db = SparqlDatabase()
    tid = db.encode_term_star(term)
    assert db.encode_term_star(term) == tid  # idempotent interning
    back = db.decode_term(tid)
    assert back is not None
    # decoded rendering re-encodes to the SAME id (round trip identity);
    # plain IRIs decode bracket-less by design, so re-encode via the
    # star-aware path only when the decoded form is a quoted triple
    if back.startswith("<<"):
        assert db.encode_term_star(back) == tid
        for leaf in ("http://t/",):
            assert leaf in back
    else:
        assert db.dictionary.lookup(back) == tid
