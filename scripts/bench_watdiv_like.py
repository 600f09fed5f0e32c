"""20-query latency suite over the employee schema — the shape of the
reference's headline WatDiv benchmark (README.md:1036: 20 queries, L/S/F/C
classes, 10M triples, "sub-millisecond to low millisecond" on CPU).

    python scripts/bench_watdiv_like.py --triples 10000000
"""
import argparse
import statistics
import sys
import time

sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parent.parent))

import torch

from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import DS, FOAF, generate_partition, plan_dataset


def build_queries(ds, db):
    e = ds.emp_base + 12345 % max(1, ds.n_employees)
    e2 = ds.emp_base + 777 % max(1, ds.n_employees)
    d = ds.dept_base + 7 % max(1, ds.n_departments)
    c = ds.city_base + 3
    for iri, tid in ((f"http://synthetic/e{e}", e),
                     (f"http://synthetic/e{e2}", e2),
                     (f"http://synthetic/d{d}", d),
                     (f"http://synthetic/c{c}", c)):
        db.dictionary.str_to_id[iri] = tid
    E, E2, D, C = (f"<http://synthetic/e{e}>", f"<http://synthetic/e{e2}>",
                   f"<http://synthetic/d{d}>", f"<http://synthetic/c{c}>")
    P = {k: f"<{v}>" for k, v in {
        "name": FOAF + "name", "home": FOAF + "workplaceHomepage",
        "sal": DS + "annual_salary", "pos": DS + "position",
        "mail": DS + "email", "age": DS + "age", "wf": DS + "worksFor",
        "loc": DS + "locatedIn",
        "label": "http://www.w3.org/2000/01/rdf-schema#label"}.items()}
    return {
        # L: linear paths
        "L1": f"SELECT ?d WHERE {{ {E} {P['wf']} ?d }}",
        "L2": f"SELECT ?c WHERE {{ {E} {P['wf']} ?d . ?d {P['loc']} ?c }}",
        "L3": f"SELECT ?c WHERE {{ {E} {P['wf']}/{P['loc']} ?c }}",
        "L4": f"SELECT ?e WHERE {{ ?e {P['wf']} {D} }}",
        "L5": f"SELECT ?e WHERE {{ ?e {P['wf']}/{P['loc']} {C} }}",
        # S: star shapes
        "S1": f"SELECT ?p ?o WHERE {{ {E} ?p ?o }}",
        "S2": f"SELECT ?n ?s WHERE {{ {E} {P['name']} ?n ; {P['sal']} ?s }}",
        "S3": (f"SELECT ?n ?s ?g ?h WHERE {{ {E} {P['name']} ?n ; "
               f"{P['sal']} ?s ; {P['pos']} ?g ; {P['home']} ?h }}"),
        "S4": (f"SELECT ?n ?s ?g ?h ?a ?m WHERE {{ {E2} {P['name']} ?n ; "
               f"{P['sal']} ?s ; {P['pos']} ?g ; {P['home']} ?h ; "
               f"{P['age']} ?a ; {P['mail']} ?m }}"),
        "S5": (f"SELECT (COUNT(*) AS ?k) WHERE {{ ?e {P['wf']} {D} ; "
               f"{P['sal']} ?s ; {P['pos']} ?g }}"),
        # F: filters / optional
        "F1": (f"SELECT (COUNT(*) AS ?k) WHERE {{ ?e {P['wf']} {D} . "
               f"?e {P['sal']} ?s . FILTER(?s > 0) }}"),
        "F2": (f"SELECT ?n WHERE {{ {E} {P['name']} ?n . "
               f"OPTIONAL {{ {E} {P['mail']} ?m }} FILTER(BOUND(?m)) }}"),
        "F3": (f"SELECT (COUNT(*) AS ?k) WHERE {{ ?e {P['wf']} {D} . "
               f"MINUS {{ ?e {P['age']} ?a }} }}"),
        "F4": f"ASK {{ {E} {P['wf']} ?d . ?d {P['loc']} ?c }}",
        "F5": (f"SELECT ?d WHERE {{ ?d ^{P['wf']} {E} }}"),
        # C: complex
        "C1": (f"SELECT (COUNT(*) AS ?k) WHERE {{ ?e {P['wf']} ?d . "
               f"?e {P['sal']} ?s . ?d {P['loc']} ?c }}"),
        "C2": (f"SELECT (COUNT(*) AS ?k) WHERE {{ ?e {P['wf']} {D} . "
               f"?e {P['sal']} ?s . ?e {P['pos']} ?g . FILTER(?s > 0) }}"),
        "C3": (f"SELECT ?g (COUNT(*) AS ?k) WHERE {{ ?e {P['pos']} ?g }} "
               f"GROUP BY ?g ORDER BY ?g"),
        "C4": (f"SELECT DISTINCT ?c WHERE {{ {D} {P['loc']} ?c }}"),
        "C5": (f"SELECT (COUNT(*) AS ?k) WHERE {{ "
               f"{{ ?e {P['wf']} {D} }} UNION {{ ?e {P['loc']} {C} }} }}"),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--triples", type=int, default=10_000_000)
    ap.add_argument("--runs", type=int, default=40)
    ap.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    args = ap.parse_args()
    ddb = DistributedDatabase(0, 1, args.device)
    ds = plan_dataset(ddb.db, args.triples)
    s, p, o = generate_partition(ds, 0, 1, 1234, args.device)
    ddb.load_shard_columns(s, p, o)
    db = ddb.db
    queries = build_queries(ds, db)
    sync = (lambda: torch.cuda.synchronize()) if args.device.startswith("cuda") \
        else (lambda: None)
    print(f"# {len(queries)} queries, {db.triple_count():,} triples, "
          f"{args.device}, p50 of {args.runs} runs")
    total_p50 = 0.0
    for name, q in queries.items():
        for _ in range(3):
            rows = db.query(q)
        sync()
        lat = []
        for _ in range(args.runs):
            t0 = time.perf_counter()
            rows = db.query(q)
            sync()
            lat.append((time.perf_counter() - t0) * 1000)
        p50 = statistics.median(lat)
        total_p50 += p50
        print(f"{name:4s} {p50:8.3f} ms  ({len(rows)} rows)")
    print(f"mean p50 across suite: {total_p50 / len(queries):.3f} ms")

    # L5 again through the columnar result API (no per-row Python lists)
    q = queries["L5"]
    cols = db.query_columns(q)
    sync()
    lat = []
    for _ in range(args.runs):
        t0 = time.perf_counter()
        cols = db.query_columns(q)
        sync()
        lat.append((time.perf_counter() - t0) * 1000)
    n = len(next(iter(cols.values()))) if cols else 0
    print(f"L5c  {statistics.median(lat):8.3f} ms  ({n} rows, columnar)")


if __name__ == "__main__":
    main()
