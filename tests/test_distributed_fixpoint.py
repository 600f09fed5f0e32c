"""Distributed semi-naive fixpoint (Δ-exchange + all-reduce termination)
must equal single-rank materialisation (SURVEY §2.10
item 4).  gloo CPU, world 2 and 4, on a program with a recursive 2-premise
rule and a NAF rule whose negative premise probes remote shards.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORKER = os.path.join(REPO, "tests", "dist_fixpoint_worker.py")


def _free_port() -> int:
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_worker(nproc: int, out_path: str) -> dict:
    if nproc == 1:
        cmd = [sys.executable, WORKER, out_path]
    else:
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={nproc}",
            "--master-addr", "127.0.0.1",
            "--master-port", str(_free_port()),
            WORKER, out_path,
        ]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                         timeout=600)
    if out.returncode != 0:
        out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                             timeout=600)
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-3000:])
    with open(out_path, "r", encoding="utf-8") as f:
        return json.load(f)


@pytest.fixture(scope="module")
def single_rank(tmp_path_factory):
    p = tmp_path_factory.mktemp("fix") / "f1.json"
    return _run_worker(1, str(p))


def test_fixpoint_world2_matches_single_rank(single_rank, tmp_path):
    r = _run_worker(2, str(tmp_path / "f2.json"))
    assert r["derived"] == single_rank["derived"]
    assert r["facts"] == single_rank["facts"]


def test_fixpoint_world4_matches_single_rank(single_rank, tmp_path):
    r = _run_worker(4, str(tmp_path / "f4.json"))
    assert r["derived"] == single_rank["derived"]
    assert r["facts"] == single_rank["facts"]


def test_fixpoint_is_nontrivial(single_rank):
    assert single_rank["derived"] > 3000
    # the NAF rule produced oneway facts
    assert len(single_rank["facts"]) > single_rank["derived"]
