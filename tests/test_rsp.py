"""RSP streaming tests (mirrors kolibrie/tests/rsp_engine_test.rs, 26 tests,
and s2r.rs inline window-firing tests :422-497).  Streams are simulated by
add_to_stream with synthetic timestamps — logical time, deterministic."""
import pytest

from kolibrie_amd.rsp import (
    CSPARQLWindow, ContentContainer, Report, ReportStrategy, RSPBuilder,
    Relation2StreamOperator, StreamOperator, Tick,
)

EX = "http://example.org/"


# ------------------------------------------------------------------- windows
def _mk_window(width, slide, strategy=ReportStrategy.ON_WINDOW_CLOSE):
    r = Report()
    r.add(strategy)
    return CSPARQLWindow(width, slide, r, Tick.TIME_DRIVEN, "w")


def test_window_firing_counts():
    w = _mk_window(10, 5)
    fired = []
    w.register_callback(lambda c: fired.append(sorted(c.items())))
    for ts in range(0, 30):
        w.add_to_window(("s", "p", f"o{ts}"), ts)
    # [0,5) fires at 5 (negative-origin saturation, ref s2r.rs `as usize`),
    # then [0,10) at 10, [5,15) at 15, [10,20) at 20, [15,25) at 25
    assert len(fired) == 5
    assert fired[0] == [("s", "p", f"o{i}") for i in range(5)]
    second = fired[1]
    assert ("s", "p", "o0") in second and ("s", "p", "o9") in second
    assert ("s", "p", "o10") not in second


def test_window_content_multiset_last_ts():
    c = ContentContainer()
    c.add(("a",), 1)
    c.add(("a",), 5)
    assert len(c) == 1
    assert dict(c.iter_with_timestamps())[("a",)] == 5


def test_window_channel_consumer():
    w = _mk_window(4, 4)
    q = w.register()
    for ts in range(0, 9):
        w.add_to_window(("e", ts), ts)
    contents = []
    while not q.empty():
        contents.append(q.get())
    assert len(contents) == 2


def test_window_flush():
    w = _mk_window(100, 100)
    fired = []
    w.register_callback(lambda c: fired.append(c))
    w.add_to_window(("x",), 1)
    w.add_to_window(("y",), 2)
    assert not fired
    w.flush()
    assert len(fired) == 1
    assert len(fired[0]) == 2


# ----------------------------------------------------------------------- r2s
def test_rstream_istream_dstream():
    r = Relation2StreamOperator(StreamOperator.ISTREAM)
    assert r.eval([(1,), (2,)], 0) == [(1,), (2,)]
    assert r.eval([(2,), (3,)], 1) == [(3,)]
    d = Relation2StreamOperator(StreamOperator.DSTREAM)
    assert d.eval([(1,), (2,)], 0) == []
    assert d.eval([(2,)], 1) == [(1,)]
    rs = Relation2StreamOperator(StreamOperator.RSTREAM)
    assert rs.eval([(1,)], 0) == [(1,)]
    assert rs.eval([(1,)], 1) == [(1,)]


# -------------------------------------------------------------------- engine
def _simple_engine(stream_type="RSTREAM", **kw):
    q = f"""
        PREFIX ex: <{EX}>
        REGISTER {stream_type} <http://out> AS
        SELECT ?s ?o
        FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 10 STEP 10]
        WHERE {{ WINDOW <http://w1> {{ ?s ex:temp ?o }} }}
    """
    outputs = []
    b = RSPBuilder().add_rsp_ql_query(q).add_consumer(outputs.append)
    for k, v in kw.items():
        getattr(b, k)(v)
    return b.build(), outputs


def test_engine_single_window_rstream():
    eng, outputs = _simple_engine()
    for ts in range(0, 25):
        eng.add_to_stream("http://s1", (f"<{EX}m{ts % 3}>", f"<{EX}temp>", f'"{ts}"'), ts)
    # windows [0,10) and [10,20) fired
    assert len(outputs) == 2
    flat = outputs[0]
    assert all(len(r) == 2 for r in flat)
    assert any(r[0] == f"{EX}m0" for r in flat)


def test_engine_istream_only_new():
    eng, outputs = _simple_engine(stream_type="ISTREAM")
    # same triple in both windows: second firing emits nothing new
    eng.add_to_stream("http://s1", (f"<{EX}a>", f"<{EX}temp>", '"1"'), 1)
    eng.add_to_stream("http://s1", (f"<{EX}a>", f"<{EX}temp>", '"1"'), 11)
    eng.add_to_stream("http://s1", (f"<{EX}a>", f"<{EX}temp>", '"1"'), 21)
    assert len(outputs) == 2
    assert len(outputs[0]) == 1   # first firing: new row
    assert outputs[1] == []       # same content -> ISTREAM empty


def test_engine_reasoning_in_window():
    q = f"""
        PREFIX ex: <{EX}>
        REGISTER RSTREAM <http://out> AS
        SELECT ?m
        FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 10 STEP 10]
        WHERE {{ WINDOW <http://w1> {{ ?m ex:alert "hot" }} }}
    """
    rules = f"""
        RULE :Hot :- CONSTRUCT {{ ?m <{EX}alert> "hot" }}
        WHERE {{ ?m <{EX}temp> ?t . FILTER(?t > 90) }} .
    """
    outputs = []
    eng = (RSPBuilder().add_rsp_ql_query(q).add_sparql_rules(rules)
           .add_consumer(outputs.append).build())
    eng.add_to_stream("http://s1", (f"<{EX}m1>", f"<{EX}temp>", '"95"'), 1)
    eng.add_to_stream("http://s1", (f"<{EX}m2>", f"<{EX}temp>", '"50"'), 2)
    eng.add_to_stream("http://s1", (f"<{EX}m3>", f"<{EX}temp>", '"99"'), 12)
    assert len(outputs) == 1
    assert sorted(r[0] for r in outputs[0]) == [f"{EX}m1"]


def test_engine_multi_window_join_wait():
    q = f"""
        PREFIX ex: <{EX}>
        REGISTER RSTREAM <http://out> AS
        SELECT ?m ?t ?l
        FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 10 STEP 10]
        FROM NAMED WINDOW <http://w2> ON STREAM <http://s2> [RANGE 10 STEP 10]
        WHERE {{
            WINDOW <http://w1> {{ ?m ex:temp ?t }}
            WINDOW <http://w2> {{ ?m ex:loc ?l }}
        }}
    """
    outputs = []
    eng = (RSPBuilder().add_rsp_ql_query(q).set_sync_policy("Wait")
           .add_consumer(outputs.append).build())
    eng.add_to_stream("http://s1", (f"<{EX}m1>", f"<{EX}temp>", '"95"'), 1)
    eng.add_to_stream("http://s1", (f"<{EX}m1>", f"<{EX}temp>", '"96"'), 11)
    assert outputs == []  # w2 never fired: Wait blocks
    eng.add_to_stream("http://s2", (f"<{EX}m1>", f"<{EX}loc>", '"lab"'), 5)
    eng.add_to_stream("http://s2", (f"<{EX}m1>", f"<{EX}loc>", '"lab"'), 15)
    assert len(outputs) == 1
    assert (f"{EX}m1", "95", "lab") in {tuple(r) for r in outputs[0]}


def test_engine_static_join():
    q = f"""
        PREFIX ex: <{EX}>
        REGISTER RSTREAM <http://out> AS
        SELECT ?m ?t
        FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 10 STEP 10]
        WHERE {{ WINDOW <http://w1> {{ ?m ex:temp ?t }} }}
    """
    outputs = []
    eng = (RSPBuilder().add_rsp_ql_query(q)
           .add_consumer(outputs.append).build())
    eng.add_static_ntriples(f'<{EX}m1> <{EX}type> "sensor" .')
    eng.add_to_stream("http://s1", (f"<{EX}m1>", f"<{EX}temp>", '"20"'), 1)
    eng.add_to_stream("http://s1", (f"<{EX}m1>", f"<{EX}temp>", '"21"'), 11)
    assert len(outputs) == 1
    assert outputs[0]


def test_engine_multithread_mode():
    from kolibrie_amd.rsp import OperationMode
    q = f"""
        PREFIX ex: <{EX}>
        REGISTER RSTREAM <http://out> AS
        SELECT ?s ?o
        FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 5 STEP 5]
        WHERE {{ WINDOW <http://w1> {{ ?s ex:p ?o }} }}
    """
    outputs = []
    eng = (RSPBuilder().add_rsp_ql_query(q)
           .set_operation_mode(OperationMode.MULTI_THREAD)
           .add_consumer(outputs.append).build())
    for ts in range(0, 12):
        eng.add_to_stream("http://s1", (f"<{EX}x>", f"<{EX}p>", f'"{ts}"'), ts)
    import time
    deadline = time.time() + 5
    while len(outputs) < 2 and time.time() < deadline:
        time.sleep(0.01)
    eng.stop()
    assert len(outputs) >= 2


def test_cross_window_sds_naive_vs_incremental():
    from kolibrie_amd.reasoning.sds import (
        Sds, WindowedTriple, incremental_sds_plus, naive_sds_plus,
    )
    from kolibrie_amd.reasoning.rule import Rule
    from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable
    from kolibrie_amd.storage.database import SparqlDatabase
    db = SparqlDatabase()
    p = db.dictionary.encode("p")
    q = db.dictionary.encode("q")
    a, b, c = (db.dictionary.encode(x) for x in "abc")
    rule = Rule(
        premise=[
            TriplePattern(Variable("x"), Constant(p), Variable("y")),
            TriplePattern(Variable("y"), Constant(p), Variable("z")),
        ],
        conclusion=[TriplePattern(Variable("x"), Constant(q), Variable("z"))],
    )
    sds = Sds()
    sds.add(WindowedTriple("w1", (a, p, b), 0), 10)   # expires at 10
    sds.add(WindowedTriple("w2", (b, p, c), 5), 10)   # expires at 15
    for ts in (1, 7, 12, 20):
        got_n = naive_sds_plus(sds, [rule], db, ts)
        got_i = incremental_sds_plus(sds, [rule], db, ts)
        assert got_n == got_i, ts
    # derived (a,q,c) lives until min(10, 15) = 10
    assert (a, q, c) in naive_sds_plus(sds, [rule], db, 7)
    assert (a, q, c) not in naive_sds_plus(sds, [rule], db, 12)


def test_bulk_ingest_matches_per_event_path():
    """K7 columnar ingest must produce the same window counts as the
    per-event host path."""
    import torch
    q = f"""PREFIX ex: <{EX}>
REGISTER RSTREAM <http://out> AS
SELECT (COUNT(*) AS ?c)
FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 10 STEP 10]
WHERE {{ WINDOW <http://w1> {{ ?m ex:temp ?v }} }}"""
    # bulk path
    out_bulk = []
    eng = RSPBuilder().add_rsp_ql_query(q).add_consumer(out_bulk.append).build()
    db = eng.store.db
    temp = db.encode_term(f"<{EX}temp>")
    n = 50
    s = torch.arange(n, dtype=torch.int32) % 7 + 10_000
    p = torch.full((n,), temp, dtype=torch.int32)
    o = torch.arange(n, dtype=torch.int32) + 20_000
    ts = (torch.arange(n, dtype=torch.int64) * 25) // n  # 0..24
    eng.add_to_stream_bulk("http://s1", s, p, o, ts)
    counts_bulk = [int(rows[0][0]) for rows in out_bulk if rows]
    # per-event oracle: same events through CSPARQLWindow + host processor
    out_host = []
    eng2 = RSPBuilder().add_rsp_ql_query(q).add_consumer(out_host.append).build()
    db2 = eng2.store.db
    for i in range(n):
        eng2.add_to_stream(
            "http://s1",
            (f"<http://sensor/{int(s[i])}>", f"<{EX}temp>", f'"{int(o[i])}"'),
            int(ts[i]))
    counts_host = [int(rows[0][0]) for rows in out_host if rows]
    assert counts_bulk == counts_host
    assert sum(counts_bulk) == sum(1 for t in ts.tolist() if t < 20)


def _timeout_engine(fallback):
    """Two-window MultiThread engine with a short Timeout policy."""
    import time
    from kolibrie_amd.parsing.ast import SyncPolicy
    from kolibrie_amd.rsp.engine import OperationMode, RSPEngine
    eng = RSPEngine(operation_mode=OperationMode.MULTI_THREAD,
                    sync_policy=SyncPolicy("Timeout", 200, fallback))
    eng.add_window("<w1>", "<s1>", 4, 4)
    eng.add_window("<w2>", "<s2>", 4, 4)
    got = []
    eng.add_consumer(lambda rows: got.append(rows))
    return eng, got


def _wait_for(pred, timeout_s=3.0):
    import time
    t0 = time.time()
    while time.time() - t0 < timeout_s:
        if pred():
            return True
        time.sleep(0.02)
    return pred()


def test_timeout_steal_policy_emits_partial():
    """ref rsp_engine_test.rs test_timeout_steal_policy: only one of two
    windows fires; after the timeout the Steal fallback emits with the
    window that did."""
    eng, got = _timeout_engine("Steal")
    for ts in range(5):
        eng.add_to_stream("<s1>", ("<a>", "<p>", f"<o{ts}>"), ts)
    # w1 fired at ts=4; w2 never does.  Expiry check runs in the workers.
    assert _wait_for(lambda: len(got) > 0)
    assert any(r for r in got)
    eng.stop()


def test_timeout_drop_policy_discards_partial():
    """ref test_timeout_drop_policy: Drop fallback discards the partial
    result set — nothing is emitted."""
    import time
    eng, got = _timeout_engine("Drop")
    for ts in range(5):
        eng.add_to_stream("<s1>", ("<a>", "<p>", f"<o{ts}>"), ts)
    time.sleep(1.0)  # > timeout; workers run the expiry check
    assert got == []
    assert eng._pending_results == {}  # partial set was dropped, not stuck
    eng.stop()


def test_timeout_wait_completes_before_expiry():
    """Both windows fire promptly: Timeout behaves like Wait (joined emit)."""
    eng, got = _timeout_engine("Drop")
    for ts in range(5):
        eng.add_to_stream("<s1>", ("<a>", "<p>", "<o>"), ts)
        eng.add_to_stream("<s2>", ("<a>", "<q>", "<u>"), ts)
    assert _wait_for(lambda: len(got) > 0)
    eng.stop()


def test_policy_timeout_fallback_grammar():
    """WITH POLICY (timeout = 5s, fallback = drop) — ref parser.rs:2737."""
    from kolibrie_amd.parsing.sparql import parse_combined_query
    q = parse_combined_query("""
        REGISTER RSTREAM <out> AS
        SELECT ?s FROM NAMED WINDOW <w1> ON STREAM <s1> [RANGE 10 STEP 5]
        WITH POLICY (timeout = 5s, fallback = drop)
        WHERE { WINDOW <w1> { ?s <p> ?o } }
    """)
    pol = q.register.windows[0].policy
    assert pol.kind == "Timeout"
    assert pol.timeout_ms == 5000
    assert pol.fallback == "Drop"


# ---- probabilistic / hybrid RSP (ref rsp_engine_test.rs:1516-1650) ----

_HYBRID_QUERY = """
    REGISTER RSTREAM <out> AS
    SELECT ?s FROM NAMED WINDOW <w> ON STREAM <s1> [RANGE 4 STEP 4]
    WHERE { WINDOW <w> { ?s <http://test/result> <http://test/yes> } }
"""
_HYBRID_RULE = """
    RULE :Hybrid PROB(provenance=hybrid, threshold=0.7) :-
    CONSTRUCT { ?s <http://test/result> <http://test/yes> }
    WHERE { ?s <http://test/input> <http://test/yes> }
"""


def _hybrid_engine():
    from kolibrie_amd.rsp.builder import RSPBuilder
    return (RSPBuilder()
            .add_rsp_ql_query(_HYBRID_QUERY)
            .add_sparql_rules(_HYBRID_RULE)
            .build())


def test_prob_annotation_selects_hybrid_path():
    eng = _hybrid_engine()
    assert eng.store.hybrid_config is not None
    assert abs(eng.store.hybrid_config.threshold - 0.7) < 1e-9


def test_rsp_deterministic_fact_dominates_probabilistic_copy():
    """Certain copy of a probabilistic event -> probability 1.0
    (ref rsp_engine_test.rs:1516)."""
    eng = _hybrid_engine()
    certain = ("<http://test/certain>", "<http://test/input>", "<http://test/yes>")
    trigger = ("<http://test/trigger>", "<http://test/input>", "<http://test/yes>")
    eng.add_to_stream("<s1>", certain, 1)
    eng.add_probabilistic_to_stream("<s1>", certain, 1, "e1", 0.2)
    eng.add_probabilistic_to_stream("<s1>", trigger, 2, "e2", 0.6)
    for entry in eng.windows.values():
        entry.window.flush()
    results = eng.latest_hybrid_results
    assert results, "hybrid window evaluation must publish results"
    assert any(r.probability is not None and abs(r.probability - 1.0) < 1e-9
               for r in results.values())


def test_rsp_probabilistic_occurrences_keep_stable_distinct_seed_ids():
    eng = _hybrid_engine()
    a = ("<http://test/a>", "<http://test/input>", "<http://test/yes>")
    b = ("<http://test/b>", "<http://test/input>", "<http://test/yes>")
    eng.add_probabilistic_to_stream("<s1>", a, 1, "occ-a", 0.4)
    eng.add_probabilistic_to_stream("<s1>", b, 2, "occ-b", 0.9)
    for entry in eng.windows.values():
        entry.window.flush()
    ids = eng._stable_seed_ids
    assert ids["occ-a"] != ids["occ-b"]
    # re-flushing does not renumber
    before = dict(ids)
    for entry in eng.windows.values():
        entry.window.flush()
    assert eng._stable_seed_ids == before


def test_rsp_probabilistic_derivation_probability():
    """Derived triple inherits its single seed's probability exactly."""
    eng = _hybrid_engine()
    t = ("<http://test/x>", "<http://test/input>", "<http://test/yes>")
    eng.add_probabilistic_to_stream("<s1>", t, 1, "seed-x", 0.8)
    for entry in eng.windows.values():
        entry.window.flush()
    results = eng.latest_hybrid_results
    assert results
    assert any(r.probability is not None and abs(r.probability - 0.8) < 1e-6
               for r in results.values())


def test_rsp_hybrid_threshold_gates_derivation():
    """Below-threshold derivations are not asserted into the window store
    (threshold=0.7; a 0.4 seed's conclusion stays out of query results)."""
    eng = _hybrid_engine()
    got = []
    eng.add_consumer(lambda rows: got.append(rows))
    lo = ("<http://test/lo>", "<http://test/input>", "<http://test/yes>")
    hi = ("<http://test/hi>", "<http://test/input>", "<http://test/yes>")
    eng.add_probabilistic_to_stream("<s1>", lo, 1, "s-lo", 0.4)
    eng.add_probabilistic_to_stream("<s1>", hi, 2, "s-hi", 0.95)
    for entry in eng.windows.values():
        entry.window.flush()
    flat = [v for rows in got for r in rows for v in r]
    assert any("hi" in v for v in flat)
    assert not any("/lo" in v for v in flat)


def test_rsp_overlapping_windows_share_one_occurrence_identity():
    """Sliding windows (RANGE 4 STEP 2): the same probabilistic occurrence
    lands in two overlapping windows but keeps ONE seed identity
    (ref rsp_engine_test.rs:1602)."""
    from kolibrie_amd.rsp.builder import RSPBuilder
    q = """
        REGISTER RSTREAM <out> AS
        SELECT ?s FROM NAMED WINDOW <w> ON STREAM <s1> [RANGE 4 STEP 2]
        WHERE { WINDOW <w> { ?s <http://test/result> <http://test/yes> } }
    """
    eng = (RSPBuilder().add_rsp_ql_query(q)
           .add_sparql_rules(_HYBRID_RULE).build())
    t = ("<http://test/x>", "<http://test/input>", "<http://test/yes>")
    eng.add_probabilistic_to_stream("<s1>", t, 3, "occ-shared", 0.6)
    # push time forward so both overlapping windows [0,4) and [2,6) fire
    for ts in range(4, 9):
        eng.add_to_stream("<s1>", ("<http://test/tick>", "<http://test/t>",
                                   f"<http://test/{ts}>"), ts)
    for entry in eng.windows.values():
        entry.window.flush()
    assert list(eng._stable_seed_ids.values()).count(
        eng._stable_seed_ids.get("occ-shared")) == 1
    assert len(eng._stable_seed_ids) == 1


def test_static_data_not_visible_in_window_query():
    """Static triples share the dictionary but never enter windows
    (ref rsp_engine_test.rs test_static_data_not_visible_in_window_query,
    rsp_engine.rs:321-326)."""
    from kolibrie_amd.rsp.builder import RSPBuilder
    q = """
        REGISTER RSTREAM <out> AS
        SELECT ?s FROM NAMED WINDOW <w> ON STREAM <s1> [RANGE 4 STEP 4]
        WHERE { WINDOW <w> { ?s <http://t/p> <http://t/o> } }
    """
    got = []
    eng = (RSPBuilder().add_rsp_ql_query(q)
           .add_static_ntriples("<http://t/static> <http://t/p> <http://t/o> .")
           .add_consumer(lambda rows: got.append(rows)).build())
    eng.add_to_stream("<s1>", ("<http://t/ev>", "<http://t/p>", "<http://t/o>"), 1)
    for entry in eng.windows.values():
        entry.window.flush()
    flat = [v for rows in got for r in rows for v in r]
    assert any("ev" in v for v in flat)
    assert not any("static" in v for v in flat)


def test_bulk_multi_window_join():
    """Two windows fed by the COLUMNAR bulk path join like the host path."""
    import torch
    from kolibrie_amd.rsp.builder import RSPBuilder
    q = """
        REGISTER RSTREAM <out> AS
        SELECT ?m ?t ?h
        FROM NAMED WINDOW <wT> ON STREAM <sT> [RANGE 10 STEP 10]
        FROM NAMED WINDOW <wH> ON STREAM <sH> [RANGE 10 STEP 10]
        WHERE {
            WINDOW <wT> { ?m <http://x/temp> ?t }
            WINDOW <wH> { ?m <http://x/hum> ?h }
        }
    """
    got = []
    eng = (RSPBuilder().add_rsp_ql_query(q)
           .add_consumer(lambda rows: got.append(rows)).build())
    db = eng.store.db
    temp = db.encode_term("<http://x/temp>")
    hum = db.encode_term("<http://x/hum>")
    m1 = db.encode_term("<http://x/m1>")
    v1 = db.encode_term('"20"')
    v2 = db.encode_term('"60"')
    mk = lambda *xs: torch.tensor(xs, dtype=torch.int32)
    ts = torch.tensor([3, 12], dtype=torch.int64)  # second event fires [0,10)
    eng.add_to_stream_bulk("<sT>", mk(m1, m1), mk(temp, temp), mk(v1, v1), ts)
    eng.add_to_stream_bulk("<sH>", mk(m1, m1), mk(hum, hum), mk(v2, v2), ts)
    flat = [r for rows in got for r in rows]
    assert any("m1" in str(r) and "20" in str(r) and "60" in str(r)
               for r in flat), flat


def test_istream_range3_step1_sliding():
    """ref rsp_ql_istream_range3_step1: RANGE 3 STEP 1 sliding window,
    ISTREAM emits only the newly-arrived solutions per firing."""
    from kolibrie_amd.rsp.builder import RSPBuilder
    q = """
        REGISTER ISTREAM <out> AS
        SELECT ?s FROM NAMED WINDOW <w> ON STREAM <s1> [RANGE 3 STEP 1]
        WHERE { WINDOW <w> { ?s <http://t/p> ?o } }
    """
    got = []
    eng = (RSPBuilder().add_rsp_ql_query(q)
           .add_consumer(lambda rows: got.append(list(rows))).build())
    for ts in range(6):
        eng.add_to_stream("<s1>", (f"<http://t/e{ts}>", "<http://t/p>",
                                   "<http://t/o>"), ts)
    flat = [v for rows in got for r in rows for v in r]
    # each event appears as an ISTREAM addition at least once, no event
    # is re-emitted as "new" twice in a row
    for i in range(4):
        assert any(f"e{i}" in v for v in flat), (i, flat)
    for rows in got:
        names = [v for r in rows for v in r]
        assert len(names) == len(set(names))


def test_istream_same_sp_diff_object_counts_both():
    """ref rsp_ql_istream_same_sp_diff_object: two events sharing (s,p)
    but different objects are distinct solutions."""
    from kolibrie_amd.rsp.builder import RSPBuilder
    q = """
        REGISTER RSTREAM <out> AS
        SELECT ?o FROM NAMED WINDOW <w> ON STREAM <s1> [RANGE 10 STEP 10]
        WHERE { WINDOW <w> { <http://t/m> <http://t/p> ?o } }
    """
    got = []
    eng = (RSPBuilder().add_rsp_ql_query(q)
           .add_consumer(lambda rows: got.append(list(rows))).build())
    eng.add_to_stream("<s1>", ("<http://t/m>", "<http://t/p>", '"a"'), 1)
    eng.add_to_stream("<s1>", ("<http://t/m>", "<http://t/p>", '"b"'), 2)
    for entry in eng.windows.values():
        entry.window.flush()
    flat = sorted(v for rows in got for r in rows for v in r)
    assert flat == ["a", "b"]


def test_single_window_emission_is_columnar():
    """The single-window emission path must run R2S over device/CPU
    COLUMNS (K10 rows_diff), not host tuple sets."""
    from kolibrie_amd.rsp.builder import RSPBuilder
    q = """
        REGISTER ISTREAM <out> AS
        SELECT ?s ?o FROM NAMED WINDOW <w> ON STREAM <s1> [RANGE 10 STEP 2]
        WHERE { WINDOW <w> { ?s <http://t/p> ?o } }
    """
    got = []
    eng = (RSPBuilder().add_rsp_ql_query(q)
           .add_consumer(lambda rows: got.append(list(rows))).build())
    for ts in range(8):
        eng.add_to_stream("<s1>", (f"<http://t/e{ts}>", "<http://t/p>",
                                   f"<http://t/o{ts % 3}>"), ts)
    assert eng.r2s.previous_cols is not None, \
        "columnar R2S path did not engage"
    assert not eng.r2s.previous, "host-set path should be unused"
    flat = [v for rows in got for r in rows for v in r]
    assert any("e0" in v for v in flat)


def test_columnar_istream_dstream_match_host_semantics():
    """eval_columns must agree with the host-set eval on random firings."""
    import torch
    from kolibrie_amd.rsp.r2s import Relation2StreamOperator, StreamOperator
    torch.manual_seed(3)
    for mode in (StreamOperator.ISTREAM, StreamOperator.DSTREAM):
        host = Relation2StreamOperator(mode)
        dev = Relation2StreamOperator(mode)
        for _ in range(6):
            n = int(torch.randint(0, 50, (1,)))
            a = torch.randint(0, 8, (n,), dtype=torch.int32)
            b = torch.randint(0, 8, (n,), dtype=torch.int32)
            host_out = host.eval([(int(x), int(y)) for x, y in zip(a, b)])
            cols = dev.eval_columns([a, b])
            dev_out = sorted(zip(cols[0].tolist(), cols[1].tolist())) \
                if cols else []
            assert sorted(set(host_out)) == dev_out, (mode, host_out, dev_out)


def test_device_window_sorted_views_match_mask_path():
    """K7 zero-copy range scoping must equal the mask fallback, and
    out-of-order batches must flip to the fallback automatically."""
    import torch
    from kolibrie_amd.rsp.ring import DeviceStreamWindow

    def run(ts_batches, force_mask=False):
        w = DeviceStreamWindow(width=10, slide=5, device="cpu")
        fired = []
        w.register_callback(
            lambda c: fired.append((c.open, c.close,
                                    c.ts.tolist(), c.s.tolist())))
        if force_mask:
            w._sorted = False
        n = 0
        for ts in ts_batches:
            k = len(ts)
            s = torch.arange(n, n + k, dtype=torch.int32)
            n += k
            w.add_batch(s, s.clone(), s.clone(),
                        torch.tensor(ts, dtype=torch.int64))
        return fired, w._sorted

    batches = [[1, 2, 4], [5, 7, 9, 11], [12, 14, 21]]
    a, sorted_a = run(batches)
    b, _ = run(batches, force_mask=True)
    assert sorted_a is True
    assert a == b
    # out-of-order batch: falls back, same results as the mask path
    ooo = [[1, 4, 2], [5, 11, 7], [21, 12]]
    c, sorted_c = run(ooo)
    assert sorted_c is False
