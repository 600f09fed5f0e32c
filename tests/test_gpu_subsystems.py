"""GPU e2e subsystem coverage beyond the kernel differentials: RSP
streaming with the columnar R2S path, updates, checkpoints, provenance
device tags — each compared against the CPU engine on identical input."""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")

EX = "http://example.org/"


@requires_gpu
def test_rsp_istream_device_matches_cpu():
    from kolibrie_amd.rsp.builder import RSPBuilder
    q = """
        REGISTER ISTREAM <out> AS
        SELECT ?s ?o FROM NAMED WINDOW <w> ON STREAM <s1> [RANGE 6 STEP 2]
        WHERE { WINDOW <w> { ?s <http://t/p> ?o } }
    """
    outs = {}
    for dev in ("cpu", "cuda:0"):
        got = []
        eng = (RSPBuilder(device=dev).add_rsp_ql_query(q)
               .add_consumer(lambda rows: got.append(list(rows))).build())
        for ts in range(10):
            eng.add_to_stream("<s1>", (f"<http://t/e{ts % 4}>",
                                       "<http://t/p>",
                                       f"<http://t/o{ts}>"), ts)
        eng.flush_windows()
        assert eng.r2s.previous_cols is not None  # columnar path engaged
        outs[dev] = got
    assert outs["cpu"] == outs["cuda:0"]


@requires_gpu
def test_update_delete_where_device_matches_cpu():
    rows = {}
    for dev in ("cpu", "cuda:0"):
        from kolibrie_amd import SparqlDatabase
        db = SparqlDatabase(device=dev)
        for i in range(500):
            db.add_triple(f"<{EX}e{i}>", f"<{EX}p>", f'"{i % 7}"')
        db.query(f'DELETE WHERE {{ ?s <{EX}p> "3" }}')
        db.query(f"""INSERT {{ ?s <{EX}q> ?o }}
                     WHERE {{ ?s <{EX}p> ?o . FILTER(?o = "5") }}""")
        rows[dev] = (db.query(f"SELECT (COUNT(*) AS ?c) WHERE "
                              f"{{ ?s <{EX}p> ?o }}"),
                     db.query(f"SELECT (COUNT(*) AS ?c) WHERE "
                              f"{{ ?s <{EX}q> ?o }}"))
    assert rows["cpu"] == rows["cuda:0"]


@requires_gpu
def test_binary_checkpoint_roundtrip_on_device(tmp_path):
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.storage import checkpoint as cp
    db = SparqlDatabase(device="cuda:0")
    for i in range(2000):
        db.add_triple(f"<{EX}e{i}>", f"<{EX}p>", f'"{i}"')
    path = str(tmp_path / "gpu_shard.npz")
    cp.save_binary(db, path, rank=0)
    db2 = SparqlDatabase(device="cuda:0")
    cp.load_binary(db2, path)
    q = f"SELECT (COUNT(*) AS ?c) WHERE {{ ?s <{EX}p> ?o }}"
    assert db2.query(q) == db.query(q)


@requires_gpu
def test_provenance_device_tags_match_cpu():
    from kolibrie_amd.reasoning.reasoner import Reasoner
    from kolibrie_amd.reasoning.rule import Rule
    from kolibrie_amd.reasoning.provenance import MinMaxProbability
    from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable
    tags = {}
    for dev in ("cpu", "cuda:0"):
        r = Reasoner(device=dev)
        e = r._i32(r.dictionary.encode("edge"))
        rc = r._i32(r.dictionary.encode("reach"))
        seeds = {}
        for i in range(12):
            r.add_fact_ids(100 + i, e, 101 + i)
            seeds[(100 + i, e, 101 + i)] = 0.5 + 0.04 * (i % 9)
        r.add_rule(Rule(
            premise=[TriplePattern(Variable("x"), Constant(e), Variable("y"))],
            conclusion=[TriplePattern(Variable("x"), Constant(rc),
                                      Variable("y"))]))
        r.add_rule(Rule(
            premise=[TriplePattern(Variable("x"), Constant(rc), Variable("y")),
                     TriplePattern(Variable("y"), Constant(rc), Variable("z"))],
            conclusion=[TriplePattern(Variable("x"), Constant(rc),
                                      Variable("z"))]))
        out = r.infer_new_facts_with_provenance(MinMaxProbability(), seeds)
        tags[dev] = {k: round(v, 6) for k, v in out.items()}
    assert tags["cpu"] == tags["cuda:0"]


@requires_gpu
def test_query_columns_device_matches_rows():
    from kolibrie_amd import SparqlDatabase
    db = SparqlDatabase(device="cuda:0")
    for i in range(3000):
        db.add_triple(f"<{EX}e{i}>", f"<{EX}p>", f'"{i % 11}"')
    q = f"SELECT ?s ?o WHERE {{ ?s <{EX}p> ?o }} ORDER BY ?s ?o LIMIT 500"
    rows = db.query(q)
    cols = db.query_columns(q)
    assert cols["s"] == [r[0] for r in rows]
    assert cols["o"] == [r[1] for r in rows]


@pytest.mark.gpu
def test_tracer_per_op_device_timing():
    """Opt-in tracer records hipEvent-timed per-operator spans on the
    device path (SURVEY §5 tracing)."""
    from kolibrie_amd import SparqlDatabase
    from kolibrie_amd.engine import tracer

    db = SparqlDatabase(device="cuda:0")
    for i in range(200):
        db.add_triple(f"<http://e/s{i}>", "<http://e/p>", f'"{i}"')
    tracer.enable()
    try:
        tracer.reset()
        db.query("SELECT (COUNT(*) AS ?c) WHERE "
                 "{ ?s <http://e/p> ?o . FILTER(?o > 10) }")
        snap = tracer.snapshot()
    finally:
        tracer.disable()
        tracer.reset()
    assert any("Scan" in k for k in snap)
    assert all(calls >= 1 and ms >= 0.0 for calls, ms in snap.values())
