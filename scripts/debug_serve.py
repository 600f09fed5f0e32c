import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import FLAGSHIP_QUERY, generate_partition, plan_dataset

dev = torch.device("cuda:0")
ddb = DistributedDatabase(0, 1, dev)
import os
N=int(os.environ.get("NTRIPLES","1000000"))
ds = plan_dataset(ddb.db, N)
s, p, o = generate_partition(ds, 0, 1, 1234, dev)
ddb.load_shard_columns(s, p, o)
db = ddb.db
for _ in range(5):
    r = db.query(FLAGSHIP_QUERY)
print("count:", r)

def timeit(f, n=200):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

print("db.query          us:", round(timeit(lambda: db.query(FLAGSHIP_QUERY)), 1))
pq = db._plan_cache[FLAGSHIP_QUERY]
from kolibrie_amd.engine.query import _run_prepared
print("_run_prepared     us:", round(timeit(lambda: _run_prepared(pq, db)), 1))
op = pq.physical
serve = getattr(op, "_chain_serve", None)
print("serve tuple:", type(serve), "graph:", type(getattr(op, "_chain_graph", None)))
if isinstance(serve, tuple):
    print("serve_chain_count us:", round(timeit(lambda: serve[0].serve_chain_count(serve[1])), 1))
