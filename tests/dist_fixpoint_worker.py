"""Worker for the distributed-fixpoint tests (NOT a test module): runs the
Δ-exchange semi-naive fixpoint on a subject-hash partitioned fact shard
and writes the final GLOBAL fact set (sorted triples) to argv[1] on rank 0.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from kolibrie_amd.parallel.dist import init_from_env, is_dist
from kolibrie_amd.parallel.dist_fixpoint import infer_fixpoint_dist
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.reasoning.seminaive import FactStore
from kolibrie_amd.storage.database import SparqlDatabase
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def _c(db, s):
    x = db.dictionary.encode(s) & 0xFFFFFFFF
    return Constant(x - 0x1_0000_0000 if x >= 0x8000_0000 else x)


def _tp(db, s, p, o):
    def t(x):
        if isinstance(x, str) and x.startswith("?"):
            return Variable(x[1:])
        return _c(db, x)
    return TriplePattern(t(s), t(p), t(o))


def build(db):
    """Edge graph: two chains + cross links + a cycle-free DAG tail."""
    edges = []
    for i in range(60):
        edges.append((f"a{i}", "edge", f"a{i+1}"))
    for i in range(40):
        edges.append((f"b{i}", "edge", f"b{i+1}"))
    for i in range(0, 40, 5):
        edges.append((f"a{i}", "edge", f"b{i}"))
    # some one-way and two-way pairs for the NAF rule
    edges.append(("x1", "edge", "x2"))
    edges.append(("x2", "edge", "x1"))
    edges.append(("y1", "edge", "y2"))
    ids = []
    for s, p, o in edges:
        ids.append((_c(db, s).id, _c(db, p).id, _c(db, o).id))
    return ids


def main():
    out_path = sys.argv[1]
    rank, world, dev = init_from_env("cpu")
    db = SparqlDatabase(device=str(dev))
    ids = build(db)

    facts = FactStore(dev)
    mine = [(s, p, o) for (s, p, o) in ids
            if world <= 1 or (s & 0xFFFFFFFF) % world == rank]
    if mine:
        t = torch.tensor(mine, dtype=torch.int32, device=dev)
        facts.add_columns(t[:, 0].contiguous(), t[:, 1].contiguous(),
                          t[:, 2].contiguous())

    rules = [
        # transitive closure (recursive, 2-premise: the Δ-exchange rule)
        Rule(premise=[_tp(db, "?x", "edge", "?y"),
                      _tp(db, "?y", "reach", "?z")],
             conclusion=[_tp(db, "?x", "reach", "?z")]),
        Rule(premise=[_tp(db, "?x", "edge", "?y")],
             conclusion=[_tp(db, "?x", "reach", "?y")]),
        # NAF: one-way edges (negative premise probes a remote shard)
        Rule(premise=[_tp(db, "?x", "edge", "?y")],
             negative_premise=[_tp(db, "?y", "edge", "?x")],
             conclusion=[_tp(db, "?x", "oneway", "?y")]),
    ]

    n = infer_fixpoint_dist(rules, facts, db, world)

    local = sorted(zip(facts.s.cpu().tolist(), facts.p.cpu().tolist(),
                       facts.o.cpu().tolist()))
    if world > 1:
        import torch.distributed as dist
        gathered = [None] * world
        dist.all_gather_object(gathered, local)
        merged = sorted(set(t for part in gathered for t in part))
        # shard disjointness: no fact may live on two ranks
        total = sum(len(part) for part in gathered)
        assert total == len(merged), (total, len(merged))
    else:
        merged = local
    if rank == 0:
        with open(out_path, "w", encoding="utf-8") as f:
            json.dump({"derived": n, "facts": merged}, f)
    if is_dist():
        import torch.distributed as dist
        dist.barrier()


if __name__ == "__main__":
    main()
