"""Sustained mixed-workload serving benchmark.

Round-robins a realistic query mix (heavy aggregate scan + selective
point lookups) against one GPU for a fixed duration and reports
throughput + latency percentiles — the serving-side view of the engine
(BASELINE metric is q/s + p50; this adds p99 under a mixed load).

    python scripts/bench_serving.py --triples 100000000 --seconds 20
"""
import argparse
import statistics
import sys
import time

sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parent.parent))

import torch

from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import (DS, FOAF, FLAGSHIP_QUERY,
                                             generate_partition, plan_dataset)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--triples", type=int, default=100_000_000)
    ap.add_argument("--seconds", type=float, default=20.0)
    ap.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    args = ap.parse_args()

    ddb = DistributedDatabase(0, 1, args.device)
    ds = plan_dataset(ddb.db, args.triples)
    s, p, o = generate_partition(ds, 0, 1, 1234, args.device)
    ddb.load_shard_columns(s, p, o)
    db = ddb.db
    emp = ds.emp_base + 424242 % max(1, ds.n_employees)
    dept = ds.dept_base + 7 % max(1, ds.n_departments)
    db.dictionary.str_to_id[f"http://synthetic/{emp}"] = emp
    db.dictionary.str_to_id[f"http://synthetic/{dept}"] = dept

    mix = [
        ("agg", FLAGSHIP_QUERY),
        ("point-star", f"PREFIX ds: <{DS}> SELECT ?p ?o WHERE {{ "
                       f"<http://synthetic/{emp}> ?p ?o }}"),
        ("point-chain", f"PREFIX ds: <{DS}> SELECT (COUNT(*) AS ?c) WHERE {{ "
                        f"?e ds:worksFor <http://synthetic/{dept}> . "
                        f"?e ds:annual_salary ?s }}"),
        ("point-name", f"PREFIX foaf: <{FOAF}> SELECT ?n WHERE {{ "
                       f"<http://synthetic/{emp}> foaf:name ?n }}"),
    ]
    for _, q in mix:
        for _ in range(3):
            db.query(q)
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()

    lat = {k: [] for k, _ in mix}
    n = 0
    t_end = time.perf_counter() + args.seconds
    t0 = time.perf_counter()
    while time.perf_counter() < t_end:
        k, q = mix[n % len(mix)]
        q0 = time.perf_counter()
        db.query(q)
        if args.device.startswith("cuda"):
            torch.cuda.synchronize()
        lat[k].append((time.perf_counter() - q0) * 1000.0)
        n += 1
    total_s = time.perf_counter() - t0

    print(f"served {n:,} queries in {total_s:.1f}s = {n / total_s:.0f} q/s "
          f"(mixed: 1/4 full-scan aggregate, 3/4 selective)")
    for k, xs in lat.items():
        xs.sort()
        p50 = statistics.median(xs)
        p99 = xs[min(len(xs) - 1, int(len(xs) * 0.99))]
        print(f"  {k:12s} n={len(xs):6d}  p50={p50:7.3f} ms  "
              f"p99={p99:7.3f} ms  max={xs[-1]:7.3f} ms")


if __name__ == "__main__":
    main()
