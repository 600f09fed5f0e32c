"""Randomized CPU-vs-GPU differential sweep.

Generates random BGP/FILTER/OPTIONAL/UNION/aggregate queries over the
synthetic employee graph and asserts the GPU engine returns exactly the
CPU oracle's results.  This is the scaled-up version of the per-kernel
differential tests: whole-engine equivalence over many plan shapes.

    python scripts/differential_sweep.py --triples 2000000 --queries 60
"""
import argparse
import os
import random
import sys

sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parent.parent))

import torch

from kolibrie_amd import SparqlDatabase
from kolibrie_amd.parallel.synthetic import DS, FOAF, plan_dataset, generate_partition

P = {
    "name": f"<{FOAF}name>",
    "homepage": f"<{FOAF}workplaceHomepage>",
    "salary": f"<{DS}annual_salary>",
    "position": f"<{DS}position>",
    "email": f"<{DS}email>",
    "age": f"<{DS}age>",
    "worksFor": f"<{DS}worksFor>",
    "locatedIn": f"<{DS}locatedIn>",
}


def gen_random_bgp(rng: random.Random) -> str:
    """Fully random COUNT(*) BGP (joins/filters/optional/minus over random
    pattern shapes incl. variable predicates) — the widest net."""
    VARS = ["?a", "?b", "?c"]
    preds = list(P.values()) + ["?pp"]

    def pat():
        s = rng.choice(VARS)
        p = rng.choice(preds)
        o = rng.choice(VARS)
        return f"{s} {p} {o}"

    # always join-connected (share ?a) — a blind cartesian at millions of
    # rows would be a fixture bug, not an engine test
    first = f"?a {rng.choice(preds)} ?b"
    parts = [first]
    if rng.random() < 0.5:
        parts.append(f"?a {rng.choice(preds)} ?c")
    pats = " . ".join(parts)
    extra = ""
    if rng.random() < 0.4:
        extra += (f" FILTER({rng.choice(VARS)} "
                  f"{rng.choice(['>', '<', '=', '!='])} {rng.randrange(9)})")
    if rng.random() < 0.3:
        extra += f" OPTIONAL {{ {pat()} }}"
    if rng.random() < 0.25:
        extra += f" MINUS {{ {pat()} }}"
    return f"SELECT (COUNT(*) AS ?c) WHERE {{ {pats}{extra} }}"


def gen_query(rng: random.Random) -> str:
    """One random query over the employee schema."""
    kind = rng.choice(["star", "chain", "filter", "optional", "union",
                      "agg", "agg2", "rows", "distinct", "values", "bind",
                      "subquery", "minus", "ask", "path", "rand", "rand"])
    if kind == "rand":
        return gen_random_bgp(rng)
    if kind == "path":
        return (f"SELECT (COUNT(*) AS ?c) WHERE {{ "
                f"?e {P['worksFor']}/{P['locatedIn']} ?city }}")
    if kind == "bind":
        return (f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e {P['worksFor']} ?d . "
                f"BIND(TRIPLE(?e, {P['worksFor']}, ?d) AS ?t) . "
                f"FILTER(isTRIPLE(?t)) }}")
    if kind == "subquery":
        return (f"SELECT (COUNT(*) AS ?c) WHERE {{ "
                f"{{ SELECT ?d WHERE {{ ?d {P['locatedIn']} ?city }} "
                f"LIMIT 200 }} ?e {P['worksFor']} ?d }}")
    if kind == "minus":
        return (f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e {P['position']} ?v . "
                f"MINUS {{ ?e {P['age']} ?a }} }}")
    if kind == "ask":
        return f"ASK {{ ?e {P['worksFor']} ?d . ?d {P['locatedIn']} ?c }}"
    if kind == "star":
        preds = rng.sample(list(P.values()), rng.randint(2, 4))
        pats = " . ".join(f"?e {p} ?v{i}" for i, p in enumerate(preds))
        return f"SELECT (COUNT(*) AS ?c) WHERE {{ {pats} }}"
    if kind == "chain":
        return (f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e {P['worksFor']} ?d . "
                f"?d {P['locatedIn']} ?city }}")
    if kind == "filter":
        cut = rng.randint(0, 200000)
        op = rng.choice([">", "<", ">=", "<="])
        return (f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e {P['salary']} ?s . "
                f"FILTER(?s {op} {cut}) }}")
    if kind == "optional":
        return (f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e {P['worksFor']} ?d . "
                f"OPTIONAL {{ ?d {P['locatedIn']} ?city }} }}")
    if kind == "union":
        a, b = rng.sample(["position", "age", "email"], 2)
        return (f"SELECT (COUNT(*) AS ?c) WHERE {{ "
                f"{{ ?e {P[a]} ?v }} UNION {{ ?e {P[b]} ?v }} }}")
    if kind == "agg":
        having = " HAVING(?c > 10)" if rng.random() < 0.4 else ""
        return (f"SELECT ?pos (COUNT(*) AS ?c) WHERE {{ "
                f"?e {P['position']} ?pos }} GROUP BY ?pos{having} "
                f"ORDER BY ?pos")
    if kind == "agg2":
        # multi-aggregate GROUP BY: exercises the native K4 hash-aggregate
        # kernel on GPU vs the torch composite on CPU
        aggs = rng.sample(["(SUM(?s) AS ?t)", "(AVG(?s) AS ?m)",
                           "(MIN(?s) AS ?lo)", "(MAX(?s) AS ?hi)",
                           "(COUNT(*) AS ?n)"], rng.randint(2, 4))
        return (f"SELECT ?pos {' '.join(aggs)} WHERE {{ "
                f"?e {P['position']} ?pos . ?e {P['salary']} ?s }} "
                f"GROUP BY ?pos ORDER BY ?pos")
    if kind == "rows":
        lim = rng.randint(5, 400)
        return (f"SELECT ?e ?city WHERE {{ ?e {P['worksFor']} ?d . "
                f"?d {P['locatedIn']} ?city }} ORDER BY ?e ?city "
                f"LIMIT {lim}")
    if kind == "distinct":
        return (f"SELECT DISTINCT ?city WHERE {{ ?d {P['locatedIn']} ?city }} "
                f"ORDER BY ?city LIMIT 50")
    # values
    return (f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e {P['position']} ?pos . "
            f"VALUES ?pos {{ <1> <2> }} }}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--triples", type=int, default=2_000_000)
    ap.add_argument("--queries", type=int, default=60)
    ap.add_argument("--seed", type=int, default=2026)
    args = ap.parse_args()
    rng = random.Random(args.seed)

    # generate ONCE on CPU: torch.Generator sequences differ per device,
    # and the differential needs bit-identical stores
    dbs = {}
    cols = None
    for dev in ["cpu"] + (["cuda:0"] if torch.cuda.is_available() else []):
        db = SparqlDatabase(device=dev)
        ds = plan_dataset(db, args.triples)
        if cols is None:
            cols = generate_partition(ds, 0, 1, 777, "cpu")
        s, p, o = (c.to(dev) for c in cols)
        db.store.insert_bulk(0, s, p, o)
        dbs[dev] = db
    if "cuda:0" not in dbs:
        print("no GPU; CPU-only smoke")
    queries = [gen_query(rng) for _ in range(args.queries)]
    bad = 0
    for i, q in enumerate(queries):
        res = {}
        for dev, db in dbs.items():
            res[dev] = sorted(map(tuple, db.query(q)))
        if "cuda:0" in res and res["cpu"] != res["cuda:0"]:
            bad += 1
            print(f"MISMATCH on query {i}:\n{q}\ncpu={res['cpu'][:5]} "
                  f"gpu={res['cuda:0'][:5]}")
        elif i % 10 == 0:
            print(f"[{i}/{len(queries)}] ok ({len(res['cpu'])} rows)")
    print(f"checked {len(queries)} queries, mismatches: {bad}")
    if bad:
        sys.exit(1)


if __name__ == "__main__":
    main()
