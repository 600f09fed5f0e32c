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
    ap.add_argument("--from-nt", action="store_true",
                    help="string ingest: N-Triples FILE -> first query")
    args = ap.parse_args()
    if args.from_nt:
        bench_from_nt(args.triples, args.device)
        return

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





def bench_from_nt(total: int, device: str):
    """String-to-query-ready: a real N-Triples FILE ->
    parse (chunk-per-thread native) -> dictionary intern -> columnar
    insert -> 4 sorted permutations -> first flagship-shaped query."""
    import os
    import tempfile
    from kolibrie_amd.ops import _native
    assert _native is not None
    path = os.path.join(tempfile.gettempdir(), f"employees_{total}.nt")
    t0 = time.perf_counter()
    lines = _native.gen_employee_nt_file(path, total)
    print(f"# generated {lines:,} NT lines "
          f"({os.path.getsize(path)/1e9:.2f} GB) in "
          f"{time.perf_counter()-t0:.1f}s (untimed)")

    db = SparqlDatabase(device=device)
    t0 = time.perf_counter()
    db.parse_ntriples_file(path)
    t_parse = time.perf_counter()
    print(f"# parse+intern+insert-call: {t_parse - t0:.2f}s")
    n = db.triple_count()
    for code in range(4):
        _ = db.store.graph_index(0).orders[code]
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    t_idx = time.perf_counter()
    q = ("PREFIX ds: <https://data.cityofchicago.org/resource/xzkq-xp2w/> "
         "SELECT (COUNT(*) AS ?c) WHERE { ?e ds:worksFor ?d . "
         "?e ds:annual_salary ?sal . ?d ds:locatedIn ?city }")
    rows = db.query(q)
    t_q = time.perf_counter()
    print(f"file->store (parse+intern+insert): {t_parse - t0:.2f}s "
          f"({n/max(1e-9, t_parse - t0)/1e6:.1f}M triples/s)")
    print(f"index build: {t_idx - t_parse:.2f}s")
    print(f"file->first-query-answered: {t_q - t0:.2f}s "
          f"({n/max(1e-9, t_q - t0)/1e6:.1f}M triples/s); "
          f"count={rows[0][0]}")
    os.unlink(path)


if __name__ == "__main__":
    main()
