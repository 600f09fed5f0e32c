"""Worker for the distributed-exchange tests (NOT a test module itself):
runs a fixed query battery at the current world size over a subject-hash
partitioned shard, asserts every rank decodes identical results, and rank 0
writes them as JSON to argv[1].

Launched by tests/test_distributed_exchange.py via torch.distributed.run
with the gloo backend (SURVEY §4 implication (d): N-rank == 1-rank).
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: F401

from kolibrie_amd.parallel.dist import init_from_env, is_dist
from kolibrie_amd.parallel.dist_engine import DistributedDatabase

EX = "http://example.org/"


def build_triples():
    tr = []
    for i in range(400):
        tr.append((f"<{EX}e{i}>", f"<{EX}worksFor>", f"<{EX}d{i % 7}>"))
        tr.append((f"<{EX}e{i}>", f"<{EX}salary>", f'"{1000 + i % 50}"'))
        if i % 3 == 0:
            tr.append((f"<{EX}e{i}>", f"<{EX}mentor>",
                       f"<{EX}e{(i * 13 + 5) % 400}>"))
    for d in range(7):
        tr.append((f"<{EX}d{d}>", f"<{EX}locatedIn>", f"<{EX}c{d % 3}>"))
    for c in range(3):
        tr.append((f"<{EX}c{c}>", f"<{EX}inCountry>", f"<{EX}n{c % 2}>"))
    return tr


QUERIES = [
    # 3-hop chain over non-subject join keys: needs exchanges on ?d and ?city
    ("chain_count",
     f"SELECT (COUNT(*) AS ?c) WHERE {{ ?e <{EX}worksFor> ?d . "
     f"?e <{EX}salary> ?s . ?d <{EX}locatedIn> ?city . "
     f"?city <{EX}inCountry> ?n }}"),
    # materializing SELECT across a shuffle
    ("select_rows",
     f"SELECT ?e ?city WHERE {{ ?e <{EX}worksFor> ?d . "
     f"?d <{EX}locatedIn> ?city }} ORDER BY ?e ?city"),
    # object-object join (neither side partitioned on the key)
    ("obj_obj_count",
     f"SELECT (COUNT(*) AS ?c) WHERE {{ ?a <{EX}worksFor> ?d . "
     f"?b <{EX}mentor> ?m . ?m <{EX}worksFor> ?d }}"),
    # distributed GROUP BY over a shuffled join
    ("group_by",
     f"SELECT ?city (COUNT(*) AS ?c) WHERE {{ ?e <{EX}worksFor> ?d . "
     f"?d <{EX}locatedIn> ?city }} GROUP BY ?city ORDER BY ?city"),
    ("group_by_sum",
     f"SELECT ?d (SUM(?s) AS ?tot) (AVG(?s) AS ?avg) WHERE {{ "
     f"?e <{EX}worksFor> ?d . ?e <{EX}salary> ?s }} GROUP BY ?d ORDER BY ?d"),
    ("distinct_cities",
     f"SELECT DISTINCT ?city WHERE {{ ?e <{EX}worksFor> ?d . "
     f"?d <{EX}locatedIn> ?city }} ORDER BY ?city"),
    ("optional",
     f"SELECT ?e ?m WHERE {{ ?e <{EX}salary> ?s . "
     f"OPTIONAL {{ ?e <{EX}mentor> ?m }} }} ORDER BY ?e ?m"),
    ("minus",
     f"SELECT ?e WHERE {{ ?e <{EX}salary> ?s "
     f"MINUS {{ ?e <{EX}mentor> ?m }} }} ORDER BY ?e"),
    ("ask", f"ASK {{ ?d <{EX}locatedIn> <{EX}c1> }}"),
    ("global_agg",
     f"SELECT (MIN(?s) AS ?lo) (MAX(?s) AS ?hi) WHERE {{ "
     f"?e <{EX}salary> ?s }}"),
]


def _plan_has_hash_exchange(op) -> bool:
    from kolibrie_amd.plan.physical import PExchange
    if isinstance(op, PExchange) and op.mode == "hash":
        return True
    for name in ("input", "left", "right"):
        child = getattr(op, name, None)
        if child is not None and hasattr(child, "__dataclass_fields__"):
            if _plan_has_hash_exchange(child):
                return True
    return False


def main():
    out_path = sys.argv[1]
    rank, world, dev = init_from_env("cpu")
    ddb = DistributedDatabase(rank, world, dev)
    ddb.add_triples_partitioned(build_triples())

    results = {}
    for name, q in QUERIES:
        results[name] = ddb.query(q)

    if world > 1:
        # the battery must actually exercise the planner-emitted shuffle
        _sel, physical, _part = ddb.prepare(QUERIES[0][1])
        import os
        if os.environ.get("KOLIBRIE_BCAST_ROWS") == "0":
            assert _plan_has_hash_exchange(physical), \
                "expected a hash PExchange in the chain_count plan"
        # every rank must hold identical decoded results
        import torch.distributed as dist
        gathered = [None] * world
        dist.all_gather_object(gathered, results)
        for peer, r in enumerate(gathered):
            assert r == results, f"rank {rank} != rank {peer}"

    if rank == 0:
        with open(out_path, "w", encoding="utf-8") as f:
            json.dump(results, f)
    if is_dist():
        import torch.distributed as dist
        dist.barrier()


if __name__ == "__main__":
    main()
