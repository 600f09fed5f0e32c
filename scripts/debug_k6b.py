import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))
from kolibrie_amd import Reasoner
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable
from kolibrie_amd.reasoning import device_fixpoint as dfx

r = Reasoner(device="cuda:0")
sub = r._i32(r.dictionary.encode("sub"))
base = 1000
depth, chains = 6, 2
for c in range(chains):
    for i in range(depth):
        n = base + c * (depth + 1) + i
        r.add_fact_ids(n, sub, n + 1)
r.add_rule(Rule(
    premise=[TriplePattern(Variable("x"), Constant(sub), Variable("y")),
             TriplePattern(Variable("y"), Constant(sub), Variable("z"))],
    conclusion=[TriplePattern(Variable("x"), Constant(sub), Variable("z"))],
))
enc = dfx._encode_rules(r.rules)
print("enc rows:", enc[0], "pred_ids:", enc[1], "adj_need:", enc[2], "disp:", enc[3])
print("facts.n", r.facts.n, "p uniq", torch.unique(r.facts.p).cpu().tolist(), "sub", sub)
res = dfx.try_device_fixpoint(r.rules, r.facts, r.db)
print("derived:", res, "facts.n now", r.facts.n)
