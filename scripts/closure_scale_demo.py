import sys, time
sys.path.insert(0, ".")
import torch
from kolibrie_amd import SparqlDatabase
db = SparqlDatabase(device="cuda:0")
n_chains, depth = 1_000_000, 10
base = 1000
node = torch.arange(n_chains * (depth + 1), dtype=torch.int32, device="cuda:0").view(n_chains, depth + 1) + base
sub = db.dictionary.encode("http://w/subClassOf")
s = node[:, :-1].reshape(-1)
o = node[:, 1:].reshape(-1)
p = torch.full_like(s, sub)
db.store.insert_bulk(0, s, p, o)
q = 'SELECT (COUNT(*) AS ?c) WHERE { ?a <http://w/subClassOf>+ ?b }'
torch.cuda.synchronize(); t0=time.perf_counter()
r1 = db.query(q); torch.cuda.synchronize(); t1=time.perf_counter()
r2 = db.query(q); torch.cuda.synchronize(); t2=time.perf_counter()
print("count:", r1[0][0], "expect", n_chains*depth*(depth+1)//2)
print(f"10M-edge p+ closure: first {t1-t0:.3f}s (materialize+count), cached {1000*(t2-t1):.2f}ms")
