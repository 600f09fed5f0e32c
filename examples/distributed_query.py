"""Planner-driven distributed execution demo: subject-hash partitioned
shards, a join whose key is NOT the partition key (the optimizer inserts
a PExchange shuffle), and a distributed GROUP BY — run it under torchrun
with 1..8 processes (gloo on CPU, RCCL on MI355X):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 examples/distributed_query.py
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from kolibrie_amd.parallel.dist import init_from_env
from kolibrie_amd.parallel.dist_engine import DistributedDatabase

EX = "http://example.org/"


def main():
    rank, world, dev = init_from_env()
    ddb = DistributedDatabase(rank, world, dev)
    triples = []
    for i in range(1000):
        triples.append((f"<{EX}e{i}>", f"<{EX}worksFor>", f"<{EX}d{i % 13}>"))
        triples.append((f"<{EX}e{i}>", f"<{EX}salary>", f'"{1000 + i % 50}"'))
    for d in range(13):
        triples.append((f"<{EX}d{d}>", f"<{EX}locatedIn>", f"<{EX}c{d % 3}>"))
    ddb.add_triples_partitioned(triples)

    rows = ddb.query(
        f"SELECT ?city (COUNT(*) AS ?n) (AVG(?s) AS ?avg) WHERE {{ "
        f"?e <{EX}worksFor> ?d . ?e <{EX}salary> ?s . "
        f"?d <{EX}locatedIn> ?city }} GROUP BY ?city ORDER BY ?city")
    if rank == 0:
        print(f"world={world} distributed GROUP BY over a shuffled join:")
        for r in rows:
            print("  ", r)
    assert len(rows) == 3


if __name__ == "__main__":
    main()
