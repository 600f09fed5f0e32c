"""Multi-process distributed correctness on CPU (gloo, world_size=2) —
the single-node analogue of the reference's differential tests (SURVEY §4
implication (d)): N-rank partitioned execution must equal 1-rank results.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_bench(nproc: int, port: int) -> dict:
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1", "--master-port", str(port),
        "bench.py", "--gpus", str(nproc), "--device", "cpu",
        "--triples", "120000", "--steps", "2", "--warmup", "1",
    ]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                         timeout=600)
    if out.returncode != 0:  # one retry: rendezvous can flake under load
        out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                             timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    for line in out.stdout.splitlines():
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output: {out.stdout[-1000:]}")


def test_distributed_count_matches_single_rank():
    r1 = _run_bench(1, _free_port())
    r2 = _run_bench(2, _free_port())
    assert r1["config"]["result_count"] == r2["config"]["result_count"]
    assert r1["config"]["result_count"] > 0


def test_all_to_all_rows_gloo():
    """Direct unit test of the shuffle primitive under gloo world=2."""
    script = r"""
import torch, torch.distributed as dist, os
from kolibrie_amd.parallel.dist import all_to_all_rows, init_from_env
rank, world, dev = init_from_env("cpu")
n = 10
vals = torch.arange(n, dtype=torch.int32) + rank * 100
dest = (torch.arange(n) % world).to(torch.int64)
out = all_to_all_rows([vals], dest)[0]
# every value v must land on rank (index % world)
expect_from_self = vals[dest == rank]
got = set(out.tolist())
for v in expect_from_self.tolist():
    assert v in got, (rank, v, got)
assert out.numel() == n, out.numel()
print("rank", rank, "ok")
"""
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node=2",
        "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
        "-m", "no_module",
    ]
    # run via -c through a wrapper file instead
    import tempfile
    with tempfile.NamedTemporaryFile("w", suffix=".py", dir=REPO,
                                     delete=False) as f:
        f.write(script)
        path = f.name
    try:
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
            path,
        ]
        out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                             timeout=300)
        assert out.returncode == 0, out.stderr[-2000:]
        assert out.stdout.count("ok") == 2
    finally:
        os.unlink(path)
