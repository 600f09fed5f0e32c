#!/usr/bin/env python3
"""Step-wise timing of the engine's own star-chain execution path."""
import sys
import time

import torch

sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parent.parent))

from kolibrie_amd.engine.bindings import Bindings
from kolibrie_amd.engine.executor import DatasetView, ExecutionContext, ExecutionEngine
from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import DS, FLAGSHIP_QUERY, generate_partition, plan_dataset
from kolibrie_amd.parsing.sparql import parse_combined_query
from kolibrie_amd.plan.lower import build_logical_plan
from kolibrie_amd.plan.optimizer import Streamertail, annotate_needed
from kolibrie_amd.engine.query import _top_needed


def main():
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    ddb = DistributedDatabase(0, 1, dev)
    ds = plan_dataset(ddb.db, 100_000_000)
    s, p, o = generate_partition(ds, 0, 1, 1234, dev)
    ddb.load_shard_columns(s, p, o)
    db = ddb.db

    cq = parse_combined_query(FLAGSHIP_QUERY)
    prefixes = dict(db.prefixes); prefixes.update(cq.prefixes)
    sel = cq.select
    stats = db.get_or_build_stats()
    logical = build_logical_plan(sel.where, db, prefixes)
    plan = Streamertail(stats).find_best_plan(logical)
    annotate_needed(plan, _top_needed(sel))
    ctx = ExecutionContext(db, DatasetView())
    eng = ExecutionEngine(ctx)

    # decompose: plan = PBindJoin(PBindJoin(scan_wf, scan_sal), scan_loc)
    inner = plan.left
    scan_wf, scan_sal, scan_loc = inner.left, inner.right, plan.right
    print("plan:", type(plan).__name__, type(inner).__name__)

    def sync():
        torch.cuda.synchronize()

    def t(fn, label, iters=10):
        for _ in range(2):
            out = fn()
        sync()
        t0 = time.perf_counter()
        for _ in range(iters):
            out = fn()
        sync()
        print(f"{label:28s} {(time.perf_counter()-t0)/iters*1000:8.3f} ms")
        return out

    u = Bindings.unit(db.device)
    step1 = t(lambda: eng.execute(scan_wf, u), "scan worksFor (PSO)")
    print("   rows:", step1.n, "cols:", step1.variables,
          "e sorted:", bool((step1.col('e')[1:] >= step1.col('e')[:-1]).all()))
    step2 = t(lambda: eng.execute(scan_sal, step1), "probe salary")
    print("   rows:", step2.n, "cols:", step2.variables)
    step3 = t(lambda: eng.execute(scan_loc, step2), "probe locatedIn")
    print("   rows:", step3.n, "cols:", step3.variables)
    t(lambda: eng.execute(plan, u), "full plan")
    from kolibrie_amd.engine.finalize import finalize_select_bindings
    t(lambda: finalize_select_bindings(sel, step3, db), "finalize")


if __name__ == "__main__":
    main()
