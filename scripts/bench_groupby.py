#!/usr/bin/env python3
"""K4 GROUP BY aggregate benchmark (VERDICT r1 item 3 'done' check):
aggregation-stage time for GROUP BY over the 100M-triple employee join,
native LDS-staged hash kernel vs the torch scatter composite.

The timed region is finalize's aggregation (rows already materialized,
results left device-resident as encoded columns) — decode-to-strings is
reported separately since returning 1M rows as strings is inherently
host-bound.

    gpurun -- 'python scripts/bench_groupby.py --triples 100000000'
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from kolibrie_amd.engine import finalize as fin
from kolibrie_amd.engine.bindings import Bindings
from kolibrie_amd.engine.executor import (DatasetView, ExecutionContext,
                                          ExecutionEngine)
from kolibrie_amd.parsing.sparql import parse_combined_query
from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import DS, generate_partition, plan_dataset
from kolibrie_amd.plan.lower import build_logical_plan
from kolibrie_amd.plan.optimizer import Streamertail, annotate_needed


def mat_rows(db, sparql, needed):
    cq = parse_combined_query(sparql)
    sel = cq.select
    logical = build_logical_plan(sel.where, db, dict(cq.prefixes))
    physical = Streamertail(db.get_or_build_stats()).find_best_plan(logical)
    annotate_needed(physical, set(needed))
    ctx = ExecutionContext(db, DatasetView())
    return sel, ExecutionEngine(ctx).execute(physical, Bindings.unit(db.device))


def time_agg(sel, rows, db, steps=10):
    torch.cuda.synchronize()
    fin.finalize_select_bindings(sel, rows, db)  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        out = fin.finalize_select_bindings(sel, rows, db)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) * 1000 / steps
    return ms, out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--triples", type=int, default=100_000_000)
    ap.add_argument("--device", default="cuda:0")
    args = ap.parse_args()
    dev = torch.device(args.device)

    ddb = DistributedDatabase(0, 1, dev)
    ds = plan_dataset(ddb.db, args.triples)
    s, p, o = generate_partition(ds, 0, 1, 1234, dev)
    ddb.load_shard_columns(s, p, o)
    db = ddb.db
    print(f"# loaded {db.triple_count():,} triples", file=sys.stderr)

    q = (f"PREFIX ds: <{DS}> SELECT ?d (COUNT(*) AS ?c) (SUM(?sal) AS ?t) "
         f"WHERE {{ ?e ds:worksFor ?d . ?e ds:annual_salary ?sal }} "
         f"GROUP BY ?d")
    q_e = (f"PREFIX ds: <{DS}> SELECT ?e (SUM(?sal) AS ?t) "
           f"WHERE {{ ?e ds:worksFor ?d . ?e ds:annual_salary ?sal }} "
           f"GROUP BY ?e")

    for name, sparql, needed in [
        ("groupby_dept", q, {"d", "sal"}),
        ("groupby_employee", q_e, {"e", "sal"}),
    ]:
        sel, rows = mat_rows(db, sparql, needed)
        print(f"# {name}: {rows.n:,} input rows", file=sys.stderr)
        ms_native, out = time_agg(sel, rows, db)
        ngroups = out.n
        # A/B: force the torch composite path
        orig = fin._native_group_aggregate
        fin._native_group_aggregate = lambda *a, **k: None
        try:
            ms_torch, out2 = time_agg(sel, rows, db)
        finally:
            fin._native_group_aggregate = orig
        assert out2.n == ngroups, (out2.n, ngroups)
        # decode cost (string materialization), once
        t0 = time.perf_counter()
        fin.decode_rows(sel, out, db)
        dec_ms = (time.perf_counter() - t0) * 1000
        print(f"{name}: rows={rows.n} groups={ngroups} "
              f"native_ms={ms_native:.3f} torch_ms={ms_torch:.3f} "
              f"speedup={ms_torch/ms_native:.2f}x decode_ms={dec_ms:.1f}")


if __name__ == "__main__":
    main()
