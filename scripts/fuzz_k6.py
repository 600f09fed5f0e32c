"""Randomized K6 differential soak: random small rule programs (within
the kernel's language: 1-2 positive premises, constant predicates, <=2
conclusions) over random fact graphs, device persistent-kernel fixpoint
vs the CPU host-oracle fixpoint — fact sets must match exactly.

    gpurun -- 'python scripts/fuzz_k6.py --programs 40'
"""
import argparse
import random
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from kolibrie_amd import Reasoner
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def random_program(rng, n_preds=4, n_rules=4):
    """Rules over predicate ids 1..n_preds and variables x,y,z."""
    def pat(pred, a, b):
        return TriplePattern(Variable(a), Constant(pred), Variable(b))

    rules = []
    for _ in range(n_rules):
        kind = rng.random()
        p1 = rng.randrange(1, n_preds + 1)
        cpred = rng.randrange(1, n_preds + 1)
        if kind < 0.35:
            # copy rule, possibly reversing and double-concluding
            concl = [pat(cpred, "x", "y")]
            if rng.random() < 0.4:
                concl.append(pat(rng.randrange(1, n_preds + 1), "y", "x"))
            rules.append(Rule(premise=[pat(p1, "x", "y")], conclusion=concl))
        else:
            p2 = rng.randrange(1, n_preds + 1)
            # one shared var, random positions
            j1 = rng.choice(["s", "o"])
            j2 = rng.choice(["s", "o"])
            v1 = ("m", "a") if j1 == "s" else ("a", "m")
            v2 = ("m", "b") if j2 == "s" else ("b", "m")
            prem = [pat(p1, *v1), pat(p2, *v2)]
            srcs = [v for v in ("a", "b", "m")]
            cs, co = rng.choice(srcs), rng.choice(srcs)
            if cs == co:
                co = "m" if cs != "m" else "a"
            rules.append(Rule(premise=prem,
                              conclusion=[pat(cpred, cs, co)]))
    return rules


def random_facts(rng, n_preds, n_nodes, n_facts):
    out = set()
    for _ in range(n_facts):
        out.add((100 + rng.randrange(n_nodes),
                 rng.randrange(1, n_preds + 1),
                 100 + rng.randrange(n_nodes)))
    return sorted(out)


def run(device, rules, facts):
    r = Reasoner(device=device)
    for s, p, o in facts:
        r.add_fact_ids(s, p, o)
    for rule in rules:
        r.add_rule(rule)
    derived = r.infer_new_facts_semi_naive()
    fs = sorted(zip(r.facts.s.cpu().tolist(), r.facts.p.cpu().tolist(),
                    r.facts.o.cpu().tolist()))
    return derived, fs, getattr(r.facts, "k6_rounds", None)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--programs", type=int, default=40)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    assert torch.cuda.is_available(), "K6 soak needs the GPU"
    rng = random.Random(args.seed)
    engaged = 0
    for i in range(args.programs):
        n_preds = rng.randrange(2, 5)
        rules = random_program(rng, n_preds, rng.randrange(1, 5))
        facts = random_facts(rng, n_preds, rng.randrange(4, 30),
                             rng.randrange(5, 120))
        d_cpu, f_cpu, _ = run("cpu", rules, facts)
        d_gpu, f_gpu, k6 = run("cuda:0", rules, facts)
        if k6 is not None:
            engaged += 1
        assert d_cpu == d_gpu, (i, d_cpu, d_gpu)
        assert f_cpu == f_gpu, (i, "fact sets differ")
        if (i + 1) % 10 == 0:
            print(f"[{i+1}/{args.programs}] ok (k6 engaged {engaged})",
                  flush=True)
    print(f"checked {args.programs} programs, k6 engaged on {engaged}, "
          f"mismatches: 0")


if __name__ == "__main__":
    main()
