import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import DS, generate_partition, plan_dataset
from kolibrie_amd.plan.physical import plan_key

dev = torch.device("cuda:0")
ddb = DistributedDatabase(0, 1, dev)
ds = plan_dataset(ddb.db, 100_000_000)
s, p, o = generate_partition(ds, 0, 1, 1234, dev)
ddb.load_shard_columns(s, p, o)
db = ddb.db
city = ds.city_base + 3
db.dictionary.str_to_id[f"http://synthetic/c{city}"] = city
q = (f"PREFIX ds: <{DS}> SELECT ?e ?sal WHERE {{ "
     f"?e ds:worksFor ?d . ?e ds:annual_salary ?sal . "
     f"?d ds:locatedIn <http://synthetic/c{city}> }}")
r = db.query(q); print("rows:", len(r))
pq = db._plan_cache[q]
print(plan_key(pq.physical)[:300])
from kolibrie_amd.engine.query import _run_prepared
from kolibrie_amd.engine.executor import ExecutionContext, ExecutionEngine
from kolibrie_amd.engine.bindings import Bindings
from kolibrie_amd.engine.finalize import finalize_select_bindings, decode_rows
def t(f, n=10):
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): r=f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1000, r
ms,_ = t(lambda: _run_prepared(pq, db)); print(f"total {ms:.3f} ms")
def ex():
    return ExecutionEngine(ExecutionContext(db, pq.view)).execute(pq.physical, Bindings.unit(db.device))
ms, rows = t(ex); print(f"execute {ms:.3f} ms rows={rows.n}")
ms, fin = t(lambda: finalize_select_bindings(pq.select, rows, db)); print(f"finalize {ms:.3f}")
ms, dec = t(lambda: decode_rows(pq.select, fin, db)); print(f"decode {ms:.3f}")
