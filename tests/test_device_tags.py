"""Device-tagged fixpoint (K6 f32 tags) vs the host provenance oracle —
differential testing over random rule/seed sets."""
import random

import pytest

from kolibrie_amd.reasoning.device_tags import (
    ScalarSemiring, infer_with_provenance_device,
)
from kolibrie_amd.reasoning.provenance import (
    AddMultProbability, ExpirationProvenance, MinMaxProbability,
)
from kolibrie_amd.reasoning.provenance_fixpoint import infer_with_provenance
from kolibrie_amd.reasoning.rule import Rule
from kolibrie_amd.storage.terms import Constant, TriplePattern, Variable


def _tp(s, p, o):
    def t(x):
        return Variable(x[1:]) if isinstance(x, str) and x.startswith("?") \
            else Constant(x)
    return TriplePattern(t(s), t(p), t(o))


P, Q, R = 100, 101, 102


def _chain_rules():
    return [
        Rule(premise=[_tp("?x", P, "?y"), _tp("?y", P, "?z")],
             conclusion=[_tp("?x", Q, "?z")]),
        Rule(premise=[_tp("?x", Q, "?y"), _tp("?y", P, "?z")],
             conclusion=[_tp("?x", Q, "?z")]),
        Rule(premise=[_tp("?x", Q, "?y")],
             conclusion=[_tp("?x", R, "?y")]),
    ]


def _close(a, b):
    return abs(a - b) < 1e-5


@pytest.mark.parametrize("seed", range(4))
def test_device_matches_host_minmax(seed):
    rng = random.Random(seed)
    seeds = {}
    for _ in range(25):
        s = rng.randint(1, 8)
        o = rng.randint(1, 8)
        seeds[(s, P, o)] = round(rng.random(), 3)
    rules = _chain_rules()
    host = infer_with_provenance(rules, dict(seeds), MinMaxProbability())
    dev = infer_with_provenance_device(rules, dict(seeds),
                                       ScalarSemiring("minmax"))
    assert set(host) == set(dev)
    for k in host:
        assert _close(host[k], dev[k]), (k, host[k], dev[k])


def test_device_matches_host_expiration():
    seeds = {(1, P, 2): 10.0, (2, P, 3): 15.0, (3, P, 4): 8.0}
    rules = _chain_rules()
    host = infer_with_provenance(rules, dict(seeds), ExpirationProvenance())
    dev = infer_with_provenance_device(rules, dict(seeds),
                                       ScalarSemiring("expiration"))
    assert set(host) == set(dev)
    for k in host:
        assert _close(host[k], dev[k]), (k, host[k], dev[k])


def test_device_tag_improvement_reenters_delta():
    # same scenario as the host test: stronger tag must propagate
    rules = [
        Rule(premise=[_tp("?x", P, "?y")], conclusion=[_tp("?x", Q, "?y")]),
        Rule(premise=[_tp("?x", Q, "?y"), _tp("?y", Q, "?z")],
             conclusion=[_tp("?x", R, "?z")]),
    ]
    seeds = {(1, P, 2): 0.3, (2, P, 3): 0.9, (1, Q, 2): 0.8}
    dev = infer_with_provenance_device(rules, seeds, ScalarSemiring("minmax"))
    assert _close(dev[(1, R, 3)], 0.8)


def test_device_naf_stratum():
    B = 103
    rules = [Rule(premise=[_tp("?x", P, "?y")],
                  negative_premise=[_tp("?x", B, "?y")],
                  conclusion=[_tp("?x", Q, "?y")])]
    seeds = {(1, P, 2): 0.9, (1, B, 2): 0.3, (3, P, 4): 0.6}
    host = infer_with_provenance(rules, dict(seeds), MinMaxProbability())
    dev = infer_with_provenance_device(rules, dict(seeds),
                                       ScalarSemiring("minmax"))
    assert _close(dev[(1, Q, 2)], host[(1, Q, 2)])  # min(0.9, 1-0.3)
    assert _close(dev[(3, Q, 4)], 0.6)


@pytest.mark.parametrize("seed", [11, 12])
def test_device_matches_host_addmult(seed):
    """AddMult (probabilistic sum-product) device tags vs host oracle over
    random multi-rule programs."""
    rng = random.Random(seed)
    rules = [
        Rule(premise=[_tp("?x", P, "?y")], conclusion=[_tp("?x", Q, "?y")]),
        Rule(premise=[_tp("?x", Q, "?y"), _tp("?y", P, "?z")],
             conclusion=[_tp("?x", R, "?z")]),
    ]
    seeds = {(rng.randint(1, 12), P, rng.randint(1, 12)):
             round(rng.uniform(0.05, 1.0), 3) for _ in range(30)}
    host = infer_with_provenance(rules, dict(seeds), AddMultProbability())
    dev = infer_with_provenance_device(rules, dict(seeds),
                                       ScalarSemiring("addmult"),
                                       device="cpu")
    assert set(host) == set(dev)
    for k in host:
        assert abs(host[k] - dev[k]) < 1e-4, k


def test_device_tags_diamond_combines_paths():
    """Two derivation paths to one fact combine per semiring (max for
    minmax; sum-of-products for addmult)."""
    rules = [Rule(premise=[_tp("?x", P, "?y")], conclusion=[_tp("?x", Q, "?y")])]
    seeds = {(1, P, 2): 0.3}
    # duplicate base path via a second rule deriving the same conclusion
    rules.append(
        Rule(premise=[_tp("?x", P, "?y")], conclusion=[_tp("?x", Q, "?y")]))
    h = infer_with_provenance(rules, dict(seeds), MinMaxProbability())
    d = infer_with_provenance_device(rules, dict(seeds),
                                     ScalarSemiring("minmax"), device="cpu")
    assert abs(h[(1, Q, 2)] - 0.3) < 1e-6
    assert abs(d[(1, Q, 2)] - 0.3) < 1e-6
