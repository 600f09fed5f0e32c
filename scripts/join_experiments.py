#!/usr/bin/env python3
"""Micro-experiments informing the GPU join cost model.

Times, on a loaded 100M-triple shard:
  1. engine auto plan (baseline)
  2. native hash_join 14M x 14M alone
  3. probe_exact with RANDOM-order probes vs SORTED probes (locality)
  4. chain plan from the big relation: worksFor -> salary probe -> dept probe
"""
import sys
import time

import torch

sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parent.parent))

from kolibrie_amd.engine.scan import scan_probe, scan_unit
from kolibrie_amd.engine.tensor_utils import pack2
from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import (
    DS, FLAGSHIP_QUERY, generate_partition, plan_dataset,
)
from kolibrie_amd import ops


def timeit(fn, iters=10, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


def main():
    dev = "cuda:0"
    ddb = DistributedDatabase(0, 1, dev)
    ds = plan_dataset(ddb.db, 100_000_000)
    s, p, o = generate_partition(ds, 0, 1, 1234, dev)
    ddb.load_shard_columns(s, p, o)
    db = ddb.db
    native = ops._native
    idx = db.store.graph_index(0)

    def i32(x):
        x &= 0xFFFFFFFF
        return x - 0x1_0000_0000 if x >= 0x8000_0000 else x

    P = {k: i32(v) for k, v in ds.pred_ids.items()}

    # base relations
    wf_s, _, wf_o = scan_unit(idx, {1: P["worksFor"]})      # (e, d) 14.2M
    sal_s, _, sal_o = scan_unit(idx, {1: P["salary"]})      # (e, sal)
    loc_s, _, loc_o = scan_unit(idx, {1: P["locatedIn"]})   # (d, city)
    print(f"worksFor={wf_s.numel():,} salary={sal_s.numel():,} "
          f"locatedIn={loc_s.numel():,}", flush=True)

    t = timeit(lambda: db.query(FLAGSHIP_QUERY))
    print(f"1. engine auto plan            {t:8.3f} ms")

    t = timeit(lambda: native.hash_join([wf_s.contiguous()], [sal_s.contiguous()]))
    print(f"2. hash_join 14Mx14M on e      {t:8.3f} ms")

    keys_rand = pack2(wf_s, torch.full_like(wf_s, P["salary"]))
    t = timeit(lambda: native.probe_exact(*idx.orders[0], keys_rand.contiguous()))
    print(f"3a. probe (e,salary) random    {t:8.3f} ms")
    keys_sorted, _ = torch.sort(keys_rand)
    t = timeit(lambda: native.probe_exact(*idx.orders[0], keys_sorted.contiguous()))
    print(f"3b. probe (e,salary) sorted    {t:8.3f} ms")

    # 4. chain from big: worksFor -> salary probe (by e) -> dept probe (by d)
    def chain():
        keys = pack2(wf_s, torch.full_like(wf_s, P["salary"]))
        li, _, sal = native.probe_exact(*idx.orders[0], keys.contiguous())
        d = wf_o[li]
        keys2 = pack2(torch.full_like(d, P["locatedIn"]), d)
        li2, _, city = native.probe_exact(*idx.orders[1], keys2.contiguous())
        return li2.numel()
    t = timeit(chain)
    print(f"4. chain worksFor->sal->dept   {t:8.3f} ms  (count={chain():,})")

    # 5. dept-first chain (current auto shape, manual)
    def chain_small_first():
        keys = pack2(torch.full_like(loc_s, P["worksFor"]), loc_s)
        li, _, e = native.probe_exact(*idx.orders[1], keys.contiguous())  # POS (p,o)->s?
        return li.numel()
    # POS order key12=(p,o) z=s: probing (worksFor, d) gives employees
    t = timeit(chain_small_first)
    print(f"5. dept->worksFor probe only   {t:8.3f} ms  (rows={chain_small_first():,})")

    # 6. count-only: skip final emit (sum counts) — upper bound for COUNT(*)
    def count_only():
        keys = pack2(torch.full_like(loc_s, P["worksFor"]), loc_s)
        li, _, e = native.probe_exact(*idx.orders[1], keys.contiguous())
        keys2 = pack2(e, torch.full_like(e, P["salary"]))
        li2, _, _ = native.probe_exact(*idx.orders[0], keys2.contiguous())
        return li2.numel()
    t = timeit(count_only)
    print(f"6. dept->wf->salary probes     {t:8.3f} ms  (count={count_only():,})")


if __name__ == "__main__":
    main()
