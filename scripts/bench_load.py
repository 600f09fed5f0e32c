"""Bulk-load benchmark: columnar insert -> all four sorted permutations
built -> first query answered (the 'query-ready' ingest rate).

    python scripts/bench_load.py --triples 100000000
"""
import argparse
import sys
import time

sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parent.parent))

import torch

from kolibrie_amd import SparqlDatabase
from kolibrie_amd.parallel.synthetic import FLAGSHIP_QUERY, plan_dataset, generate_partition


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--triples", type=int, default=100_000_000)
    ap.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    args = ap.parse_args()

    db = SparqlDatabase(device=args.device)
    ds = plan_dataset(db, args.triples)
    s, p, o = generate_partition(ds, 0, 1, 7, args.device)
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()

    t0 = time.perf_counter()
    db.store.insert_bulk(0, s, p, o)
    n = db.triple_count()                       # forces dedup/commit
    t_ins = time.perf_counter()
    db.store.graph_index(0)                     # builds remaining orders
    for code in range(4):
        _ = db.store.graph_index(0).orders[code]
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()
    t_idx = time.perf_counter()
    rows = db.query(FLAGSHIP_QUERY)
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()
    t_q = time.perf_counter()

    print(f"triples: {n:,}")
    print(f"insert+dedup: {t_ins - t0:.2f}s  "
          f"({n / max(1e-9, t_ins - t0) / 1e6:.0f}M triples/s)")
    print(f"4-permutation index build: {t_idx - t_ins:.2f}s")
    print(f"query-ready total: {t_q - t0:.2f}s "
          f"({n / max(1e-9, t_q - t0) / 1e6:.0f}M triples/s); "
          f"first query count={rows[0][0]}")


if __name__ == "__main__":
    main()
