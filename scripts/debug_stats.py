import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import generate_partition, plan_dataset
from kolibrie_amd.plan.stats import DatabaseStats

ddb = DistributedDatabase(0, 1, "cuda:0")
ds = plan_dataset(ddb.db, 100_000_000)
s, p, o = generate_partition(ds, 0, 1, 1234, "cuda:0")
ddb.load_shard_columns(s, p, o)
db = ddb.db
db.store.commit_all(); db.store.graph_index(0)
torch.cuda.synchronize()
t0 = time.perf_counter()
st = DatabaseStats.gather(db)
torch.cuda.synchronize()
print(f"stats gather at 100M: {time.perf_counter()-t0:.3f}s "
      f"(preds={len(st.pred_count)})")
