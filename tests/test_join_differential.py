"""Differential join testing: the vectorized merge join must agree with a
brute-force compatibility nested loop on every row shape (ref
engine.rs:41-265 hash_join_agrees_with_the_nested_loop_on_every_row_shape —
adversarial shapes: unbound keys, conflicts, cartesian)."""
import random

import pytest
import torch

from kolibrie_amd.engine.bindings import Bindings
from kolibrie_amd.engine.executor import join_bindings
from kolibrie_amd.storage.terms import UNBOUND


def _oracle_join(left_rows, right_rows, lvars, rvars):
    """Brute-force SPARQL-compatible join over dict rows."""
    out = []
    for lr in left_rows:
        for rr in right_rows:
            ok = True
            for v in lr:
                if v in rr and lr[v] is not None and rr[v] is not None \
                        and lr[v] != rr[v]:
                    ok = False
                    break
            if ok:
                merged = {}
                for v in set(lr) | set(rr):
                    a, b = lr.get(v), rr.get(v)
                    merged[v] = a if a is not None else b
                out.append(merged)
    return out


def _to_bindings(rows, vars_):
    full = [{v: r.get(v) for v in vars_} for r in rows]
    b = Bindings.from_dicts(full, "cpu") if full else Bindings.empty("cpu", vars_)
    # from_dicts drops all-None columns? ensure all vars exist
    for v in vars_:
        if not b.has(v):
            b.cols[v] = torch.full((b.n,), UNBOUND, dtype=torch.int32)
    return b


def _normalize(rows):
    return sorted(
        tuple(sorted((k, v) for k, v in r.items() if v is not None))
        for r in rows
    )


@pytest.mark.parametrize("seed", range(12))
def test_join_agrees_with_nested_loop(seed):
    rng = random.Random(seed)
    lvars = ["a", "b"]
    rvars = ["b", "c"] if seed % 3 else ["a", "b", "c"]
    def gen(vars_, n):
        rows = []
        for _ in range(n):
            r = {}
            for v in vars_:
                roll = rng.random()
                if roll < 0.2:
                    r[v] = None  # unbound
                else:
                    r[v] = rng.randint(1, 4)
            rows.append(r)
        return rows
    left = gen(lvars, rng.randint(0, 8))
    right = gen(rvars, rng.randint(0, 8))
    lb = _to_bindings(left, lvars)
    rb = _to_bindings(right, rvars)
    got = join_bindings(lb, rb).to_dicts()
    want = _oracle_join(left, right, lvars, rvars)
    assert _normalize(got) == _normalize(want)


def test_join_cartesian_no_shared():
    l = Bindings.from_dicts([{"a": 1}, {"a": 2}], "cpu")
    r = Bindings.from_dicts([{"b": 7}, {"b": 8}, {"b": 9}], "cpu")
    out = join_bindings(l, r)
    assert out.n == 6


def test_join_multiset_duplicates():
    l = Bindings.from_dicts([{"k": 1}, {"k": 1}], "cpu")
    r = Bindings.from_dicts([{"k": 1, "v": 5}, {"k": 1, "v": 6}], "cpu")
    out = join_bindings(l, r)
    assert out.n == 4  # 2x2 matches preserved
