"""Examples stay runnable (smoke)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

EXAMPLES = [
    "examples/sparql_basics.py",
    "examples/rdf_star.py",
    "examples/streaming_rsp.py",
    "examples/hybrid_probability.py",
    "examples/mqtt_ingestion.py",
    "examples/citybench_traffic.py",
    "examples/distributed_query.py",
]


@pytest.mark.parametrize("path", EXAMPLES)
def test_example_runs(path):
    out = subprocess.run([sys.executable, path], cwd=REPO,
                         capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-1500:]
    assert out.stdout.strip()


def test_gen_data_roundtrip(tmp_path):
    out_file = tmp_path / "emp.rdf"
    r = subprocess.run(
        [sys.executable, "examples/synthetic_data/gen_data.py",
         "--employees", "50", "--out", str(out_file)],
        cwd=REPO, capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr[-800:]
    from kolibrie_amd import SparqlDatabase
    db = SparqlDatabase()
    db.parse_rdf_from_file(str(out_file))
    assert db.triple_count() == 200  # 4 triples per employee
    rows = db.query("""
        PREFIX ds: <https://data.cityofchicago.org/resource/xzkq-xp2w/>
        SELECT (COUNT(*) AS ?c) WHERE { ?e ds:annual_salary ?s }""")
    assert rows == [["50"]]


def test_fraud_detection_example():
    from examples.fraud_detection import main
    verdicts = main()
    assert len(verdicts) == 10
