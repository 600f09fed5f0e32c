#!/usr/bin/env python3
"""BASELINE config 4: Datalog transitive-closure (TBox subClassOf)
semi-naive fixpoint on GPU, 10M-edge graph; plus the deep-taxonomy shape
(depth 10/100/1K/10K sub-second).

Graph shape: a forest of subclass chains (TBox-like), so the closure stays
bounded: n_chains chains of depth d => edges = n_chains*d, closure size =
n_chains * d*(d+1)/2.
"""
import argparse
import sys
import time

import torch

sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parent.parent))

from kolibrie_amd import Reasoner
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def build_deep_taxonomy(depth: int, device: str) -> Reasoner:
    """EYE deep-taxonomy shape (ref deep_taxonomy.rs): a subclass chain of
    `depth` levels, one typed instance, and the type-propagation rule
    (X type C), (C subClassOf D) -> (X type D)."""
    r = Reasoner(device=device)
    sub = r._i32(r.dictionary.encode("rdfs:subClassOf"))
    typ = r._i32(r.dictionary.encode("rdf:type"))
    base = 1000
    import torch as _t
    node = _t.arange(depth + 1, dtype=_t.int32) + base
    s = node[:-1]
    o = node[1:]
    p = _t.full_like(s, sub)
    dev = _t.device(device)
    r.add_fact_columns(s.to(dev), p.to(dev), o.to(dev))
    r.add_fact_ids(1, typ, base)  # instance 1 at level 0
    r.add_rule(Rule(
        premise=[TriplePattern(Variable("x"), Constant(typ), Variable("c")),
                 TriplePattern(Variable("c"), Constant(sub), Variable("d"))],
        conclusion=[TriplePattern(Variable("x"), Constant(typ), Variable("d"))],
    ))
    return r


def build_reasoner(n_chains: int, depth: int, device: str) -> Reasoner:
    r = Reasoner(device=device)
    sub = r.dictionary.encode("rdfs:subClassOf")
    sub_i32 = r._i32(sub)
    base = 1000
    # chain c: nodes base + c*(depth+1) + i
    node = torch.arange(n_chains * (depth + 1), dtype=torch.int32)
    node = node.view(n_chains, depth + 1) + base
    s = node[:, :-1].reshape(-1)
    o = node[:, 1:].reshape(-1)
    p = torch.full_like(s, sub_i32)
    dev = torch.device(device)
    r.add_fact_columns(s.to(dev), p.to(dev), o.to(dev))
    r.add_rule(Rule(
        premise=[TriplePattern(Variable("x"), Constant(sub_i32), Variable("y")),
                 TriplePattern(Variable("y"), Constant(sub_i32), Variable("z"))],
        conclusion=[TriplePattern(Variable("x"), Constant(sub_i32), Variable("z"))],
    ))
    return r


def run(n_chains: int, depth: int, device: str) -> dict:
    r = build_reasoner(n_chains, depth, device)
    n_edges = n_chains * depth
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    derived = r.infer_new_facts_semi_naive()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    expect = n_chains * depth * (depth + 1) // 2 - n_edges
    assert derived == expect, (derived, expect)
    return {"edges": n_edges, "derived": derived, "seconds": dt,
            "edges_per_s": (n_edges + derived) / dt}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--huge", action="store_true",
                    help="add a 100M-edge closure config (GPU capacity demo)")
    args = ap.parse_args()

    print("== deep taxonomy (type propagation, BASELINE.md item 2) ==",
          flush=True)
    for depth in ([10, 100, 1000] if args.quick else [10, 100, 1000, 10000]):
        # untimed warmup run at the same depth: ROCm sort kernels compile
        # per (algorithm, size-class) on first use on a fresh box — that
        # one-time cost is not per-query latency
        rw = build_deep_taxonomy(depth, args.device)
        rw.infer_new_facts_semi_naive()
        r = build_deep_taxonomy(depth, args.device)
        if args.device.startswith("cuda"):
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        derived = r.infer_new_facts_semi_naive()
        if args.device.startswith("cuda"):
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        assert derived == depth, (derived, depth)
        print(f"depth {depth:6d}: {dt*1000:9.1f} ms "
              f"({derived:,} derived)", flush=True)

    print("== 10M-edge transitive closure (BASELINE config 4) ==", flush=True)
    confs = [(100_000, 10), (1_000_000, 10)] if not args.quick else [(10_000, 10)]
    if args.huge:
        confs.append((10_000_000, 10))   # 100M edges -> 550M total facts
    for n_chains, depth in confs:
        res = run(n_chains, depth, args.device)
        print(f"chains {n_chains:,} depth {depth}: edges {res['edges']:,} "
              f"derived {res['derived']:,} in {res['seconds']:.3f}s "
              f"({res['edges_per_s']/1e6:.1f}M facts/s)", flush=True)


if __name__ == "__main__":
    main()
