"""General-vtree SDD parity corpus: random formulas
compiled through the OBDD manager (right-linear special case), the vtree
manager with a right-linear vtree, and with a balanced vtree must agree
with brute-force weighted model counting."""
import itertools
import random

import pytest

from kolibrie_amd.reasoning.sdd import SddManager
from kolibrie_amd.reasoning.sdd_vtree import VtreeSddManager


def _rand_formula(rng, n_vars, depth):
    """Formula AST: ('var', i) | ('not', f) | ('and'|'or', f, g)."""
    if depth == 0 or rng.random() < 0.3:
        return ("var", rng.randrange(n_vars))
    op = rng.choice(["and", "or", "not"])
    if op == "not":
        return ("not", _rand_formula(rng, n_vars, depth - 1))
    return (op, _rand_formula(rng, n_vars, depth - 1),
            _rand_formula(rng, n_vars, depth - 1))


def _compile(m, f):
    if f[0] == "var":
        return m.literal(f[1] + 1, True)
    if f[0] == "not":
        return m.negate(_compile(m, f[1]))
    a = _compile(m, f[1])
    b = _compile(m, f[2])
    return m.conjoin(a, b) if f[0] == "and" else m.disjoin(a, b)


def _eval(f, assign):
    if f[0] == "var":
        return assign[f[1]]
    if f[0] == "not":
        return not _eval(f[1], assign)
    if f[0] == "and":
        return _eval(f[1], assign) and _eval(f[2], assign)
    return _eval(f[1], assign) or _eval(f[2], assign)


def _brute_wmc(f, n_vars, weights):
    total = 0.0
    for bits in itertools.product([False, True], repeat=n_vars):
        if _eval(f, list(bits)):
            w = 1.0
            for i, b in enumerate(bits):
                w *= weights[i] if b else (1.0 - weights[i])
            total += w
    return total


@pytest.mark.parametrize("seed", range(8))
def test_vtree_sdd_wmc_parity(seed):
    rng = random.Random(seed)
    n_vars = rng.randrange(3, 9)
    weights = [round(rng.uniform(0.05, 0.95), 3) for _ in range(n_vars)]
    f = _rand_formula(rng, n_vars, 5)
    expect = _brute_wmc(f, n_vars, weights)
    managers = [
        SddManager(),
        VtreeSddManager("right", list(range(1, n_vars + 1))),
        VtreeSddManager("balanced", list(range(1, n_vars + 1))),
    ]
    for m in managers:
        for i in range(n_vars):
            m.declare_var(i + 1, pos_weight=weights[i])
        node = _compile(m, f)
        got = m.wmc(node)
        assert got == pytest.approx(expect, rel=1e-9), type(m).__name__


def test_vtree_incremental_declaration():
    """Vars declared on the fly (right-spine growth) must not invalidate
    existing nodes."""
    m = VtreeSddManager()
    a = m.literal(1, True)
    m.declare_var(1, pos_weight=0.5)
    b = m.literal(2, True)
    m.declare_var(2, pos_weight=0.25)
    ab = m.conjoin(a, b)
    assert m.wmc(ab) == pytest.approx(0.125)
    c = m.literal(3, True)
    m.declare_var(3, pos_weight=0.5)
    f = m.disjoin(ab, c)
    # P(ab or c) = 0.125 + 0.5 - 0.0625
    assert m.wmc(f) == pytest.approx(0.5625)


def test_balanced_vtree_beats_right_linear_on_blockwise_formula():
    """The classic structured case: OR of disjoint AND-blocks.  A vtree
    aligned with the blocks (balanced) compiles smaller than the
    right-linear chain."""
    n_blocks = 8
    vars_ = list(range(1, 2 * n_blocks + 1))

    def build(m):
        total = m.false_node()
        for i in range(n_blocks):
            a = m.literal(2 * i + 1, True)
            b = m.literal(2 * i + 2, True)
            total = m.disjoin(total, m.conjoin(a, b))
        return total

    mb = VtreeSddManager("balanced", vars_)
    mr = VtreeSddManager("right", vars_)
    for v in vars_:
        mb.declare_var(v, 0.5)
        mr.declare_var(v, 0.5)
    nb = build(mb)
    nr = build(mr)
    assert mb.wmc(nb) == pytest.approx(mr.wmc(nr))
    assert mb.node_count() < mr.node_count(), \
        (mb.node_count(), mr.node_count())


def test_vtree_budget_try_ops():
    from kolibrie_amd.reasoning.sdd import SddOperationBudget
    m = VtreeSddManager("balanced", list(range(1, 17)))
    for v in range(1, 17):
        m.declare_var(v, 0.5)
    # parity function blows up any SDD: the node cap must trip
    f = m.literal(1, True)
    budget = SddOperationBudget(max_nodes=m.node_count() + 5)
    out = None
    for v in range(2, 17):
        out = m.try_apply("xor", f, m.literal(v, True), budget)
        if out is None:
            break
        f = out
    assert out is None


def test_vtree_models_match_truth_table():
    rng = random.Random(42)
    n_vars = 4
    f = _rand_formula(rng, n_vars, 4)
    m = VtreeSddManager("balanced", list(range(1, n_vars + 1)))
    node = _compile(m, f)
    sat = set()
    for bits in itertools.product([False, True], repeat=n_vars):
        if _eval(f, list(bits)):
            sat.add(bits)
    got = set()
    for partial in m.models(node):
        # expand unconstrained vars
        free = [i for i in range(n_vars) if (i + 1) not in partial]
        for bits in itertools.product([False, True], repeat=len(free)):
            full = list(range(n_vars))
            for i in range(n_vars):
                if (i + 1) in partial:
                    full[i] = partial[i + 1]
            for j, i in enumerate(free):
                full[i] = bits[j]
            got.add(tuple(full))
    assert got == sat
