#!/usr/bin/env python3
"""Selective point-query latency over 100M triples (the WatDiv-style shape
behind the reference's "sub-millisecond to low millisecond" claim —
BASELINE.md item 1): per-pattern index probes with tiny results."""
import sys
import time

import torch

sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parent.parent))

from kolibrie_amd.parallel.dist_engine import DistributedDatabase
from kolibrie_amd.parallel.synthetic import DS, generate_partition, plan_dataset


def main():
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 100_000_000
    ddb = DistributedDatabase(0, 1, dev)
    ds = plan_dataset(ddb.db, n)
    s, p, o = generate_partition(ds, 0, 1, 1234, dev)
    ddb.load_shard_columns(s, p, o)
    db = ddb.db
    emp = ds.emp_base + 12345
    dept = ds.dept_base + 77

    queries = {
        "S1 subject star (1 employee, all attrs)":
            f"PREFIX ds: <{DS}> SELECT ?p ?o WHERE {{ <#e> ?p ?o }}"
            .replace("<#e>", f"<http://synthetic/{emp}>"),
        "L1 employee lookup + salary":
            f"""PREFIX ds: <{DS}> SELECT ?sal WHERE {{
                <#e> ds:annual_salary ?sal }}""".replace("<#e>", f"<http://synthetic/{emp}>"),
        "F1 dept members via join":
            f"""PREFIX ds: <{DS}>
            SELECT (COUNT(*) AS ?c) WHERE {{
                ?e ds:worksFor <#d> . ?e ds:annual_salary ?sal }}"""
            .replace("<#d>", f"<http://synthetic/{dept}>"),
        "S5 star: 5 attributes of 1 employee":
            f"""PREFIX ds: <{DS}> PREFIX foaf: <http://xmlns.com/foaf/0.1/>
            SELECT ?n ?sal ?pos ?hp ?age WHERE {{
                <#e> foaf:name ?n ; ds:annual_salary ?sal ;
                     ds:position ?pos ; foaf:workplaceHomepage ?hp ;
                     ds:age ?age }}""".replace("<#e>", f"<http://synthetic/{emp}>"),
        "C2 complex: dept chain + filter":
            f"""PREFIX ds: <{DS}>
            SELECT (COUNT(*) AS ?c) WHERE {{
                ?e ds:worksFor <#d> . ?e ds:annual_salary ?sal .
                ?e ds:position ?pos . FILTER(?sal > 0) }}"""
            .replace("<#d>", f"<http://synthetic/{dept}>"),
    }
    # synthetic entity ids are not interned as IRIs; bind them directly
    db.dictionary.str_to_id[f"http://synthetic/{emp}"] = emp
    db.dictionary.str_to_id[f"http://synthetic/{dept}"] = dept

    for name, q in queries.items():
        for _ in range(3):
            rows = db.query(q)
        if dev.startswith("cuda"):
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(50):
            rows = db.query(q)
        if dev.startswith("cuda"):
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 50 * 1000
        print(f"{name:42s} {dt:8.3f} ms  ({len(rows)} rows)")


if __name__ == "__main__":
    main()
