"""cProfile of non-COUNT cached-plan queries (the torch-glue host path):
where do the ~0.3 ms/query go on S/C-class shapes?"""
import cProfile, io, pstats, sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import generate_partition, plan_dataset
sys.path.insert(0, str(Path(__file__).resolve().parent))
from bench_watdiv_like import build_queries

dev = "cuda:0" if torch.cuda.is_available() else "cpu"
ddb = DistributedDatabase(0, 1, dev)
ds = plan_dataset(ddb.db, 20_000_000)
s, p, o = generate_partition(ds, 0, 1, 1234, dev)
ddb.load_shard_columns(s, p, o)
db = ddb.db
queries = build_queries(ds, db)
for name in ("S3", "C1", "F2"):
    q = queries[name]
    for _ in range(5):
        db.query(q)
    torch.cuda.synchronize() if dev != "cpu" else None
    t0 = time.perf_counter()
    pr = cProfile.Profile()
    pr.enable()
    for _ in range(200):
        db.query(q)
    pr.disable()
    dt = (time.perf_counter() - t0) / 200 * 1000
    out = io.StringIO()
    ps = pstats.Stats(pr, stream=out).sort_stats("cumulative")
    ps.print_stats(14)
    print(f"==== {name}: {dt:.3f} ms/query")
    print("\n".join(out.getvalue().splitlines()[4:22]))
