import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import DS, generate_partition, plan_dataset

dev = torch.device("cuda:0")
ddb = DistributedDatabase(0, 1, dev)
ds = plan_dataset(ddb.db, 100_000_000)
s, p, o = generate_partition(ds, 0, 1, 1234, dev)
ddb.load_shard_columns(s, p, o)
db = ddb.db
c = ds.city_base + 3
db.dictionary.str_to_id[f"http://synthetic/c{c}"] = c
q = (f"SELECT ?e WHERE {{ ?e <{DS}worksFor>/<{DS}locatedIn> "
     f"<http://synthetic/c{c}> }}")
rows = db.query(q)
print("rows:", len(rows))

from kolibrie_amd.engine.query import _prepare_select, _run_prepared
from kolibrie_amd.engine.finalize import finalize_select_bindings, decode_rows
from kolibrie_amd.engine.executor import ExecutionContext, ExecutionEngine
from kolibrie_amd.engine.bindings import Bindings
pq = db._plan_cache[q]

def t(f, n=20):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): r = f()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/n*1000, r

ms, _ = t(lambda: _run_prepared(pq, db)); print(f"total        {ms:.3f} ms")
def ex():
    ctx = ExecutionContext(db, pq.view)
    return ExecutionEngine(ctx).execute(pq.physical, Bindings.unit(db.device))
ms, rws = t(ex); print(f"execute      {ms:.3f} ms  rows={rws.n}")
ms, fin = t(lambda: finalize_select_bindings(pq.select, rws, db)); print(f"finalize     {ms:.3f} ms")
ms, dec = t(lambda: decode_rows(pq.select, fin, db)); print(f"decode       {ms:.3f} ms")
print("plan:", type(pq.physical).__name__)
from kolibrie_amd.plan.physical import plan_key
print(plan_key(pq.physical)[:200])
