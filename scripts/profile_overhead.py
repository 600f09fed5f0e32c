"""Host-side (Python) overhead profile of the flagship query path.

The fused chain-count kernel is ~630us of the 767us step at 100M triples;
this cProfiles the remaining host work (plan-cache hit -> executor walk ->
launch -> sync -> finalize/decode) to find what to shave for strong
scaling (at 8 GPUs the per-rank kernel shrinks 8x but host time doesn't).
"""
import argparse
import cProfile
import io
import os
import pstats
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from kolibrie_amd import SparqlDatabase
from kolibrie_amd.engine.query import execute_query
from kolibrie_amd.parallel.synthetic import FLAGSHIP_QUERY, plan_dataset, generate_partition


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--triples", type=int, default=20_000_000)
    ap.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    ap.add_argument("--iters", type=int, default=100)
    args = ap.parse_args()

    db = SparqlDatabase(device=args.device)
    ds = plan_dataset(db, args.triples)
    s, p, o = generate_partition(ds, 0, 1, 1234, args.device)
    db.store.insert_bulk(0, s, p, o)

    # warm plan + region caches
    for _ in range(3):
        execute_query(FLAGSHIP_QUERY, db)
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()

    t0 = time.perf_counter()
    pr = cProfile.Profile()
    pr.enable()
    for _ in range(args.iters):
        execute_query(FLAGSHIP_QUERY, db)
    pr.disable()
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters * 1e3

    buf = io.StringIO()
    st = pstats.Stats(pr, stream=buf)
    st.sort_stats("cumulative").print_stats(40)
    print(buf.getvalue())
    print(f"avg end-to-end per query: {dt:.3f} ms over {args.iters} iters")


if __name__ == "__main__":
    main()
