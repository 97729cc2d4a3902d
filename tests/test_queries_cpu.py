"""Oracle vs brute-force pin + golden fixtures (DESIGN.md §4 pins 2-3)."""
import json
import os

import numpy as np
import pytest

import wukong_amd as wk
from wukong_amd import Plan, queries as Q
from tests.oracle_util import OracleCtx, OracleExecutor, sort_rows

GOLD = os.path.join(os.path.dirname(__file__), "golden", "lubm4_golden.json")


def fnv1a(arr):
    h = np.uint64(1469598103934665603)
    p = np.uint64(1099511628211)
    for b in np.asarray(arr, dtype=np.uint32).tobytes():
        h = np.uint64((int(h) ^ b) * int(p) & 0xFFFFFFFFFFFFFFFF)
    return int(h)


def fnv1a_fast(arr):
    # order-independent checksum of the sorted row set, cheap in numpy
    a = sort_rows(np.asarray(arr, dtype=np.uint32))
    import hashlib
    return hashlib.sha256(a.tobytes()).hexdigest()[:16]


@pytest.mark.parametrize("name", list(Q.ALL))
def test_oracle_vs_brute(name, lubm4, oracle4):
    plan = Q.ALL[name]
    a = sort_rows(oracle4.run_query(plan))
    b = sort_rows(oracle4.brute_query(plan))
    assert a.shape == b.shape
    assert np.array_equal(a, b)


def test_q3_empty(oracle4):
    assert oracle4.run_query(Q.Q3).shape[0] == 0


def test_mt_slicing_equal(oracle4):
    for name in ("q1", "q2", "q7"):
        a = sort_rows(oracle4.run_query(Q.ALL[name], mt=1))
        b = sort_rows(oracle4.run_query(Q.ALL[name], mt=4))
        assert np.array_equal(a, b), name


def test_distinct_limit_offset(oracle4):
    base = Q.Q2
    # DISTINCT on course (X) only
    p = Plan(base.patterns, base.nvars, [Q.X], distinct=True)
    a = oracle4.run_query(p)
    assert a.shape[0] == len(np.unique(a[:, 0]))
    # LIMIT
    p2 = Plan(base.patterns, base.nvars, base.required_vars, limit=10)
    assert oracle4.run_query(p2).shape[0] == 10
    # OFFSET beyond end
    n = oracle4.run_query(base).shape[0]
    p3 = Plan(base.patterns, base.nvars, base.required_vars, offset=n + 5)
    assert oracle4.run_query(p3).shape[0] == 0


def test_step_executor_matches_run(oracle4):
    for name in ("q1", "q5", "q7"):
        plan = Q.ALL[name]
        ex = OracleExecutor(oracle4, plan)
        for _ in plan.patterns:
            ex.step()
        got = sort_rows(ex.finalize())
        want = sort_rows(oracle4.run_query(plan))
        assert np.array_equal(got, want), name


def test_golden_fixture(oracle4):
    """Committed fixture pins the oracle against drift (make_golden.py)."""
    if not os.path.exists(GOLD):
        pytest.skip("golden fixture not generated yet")
    with open(GOLD) as f:
        gold = json.load(f)
    plans = dict(Q.ALL)
    plans.update(Q.versatile_plans(oracle4))
    for name, rec in gold["queries"].items():
        t = oracle4.run_query(plans[name])
        assert t.shape[0] == rec["rows"], name
        assert fnv1a_fast(t) == rec["sha"], name


def test_oracle_mt_slicing_equal(lubm2):
    """cpu_baseline runs the oracle with mt=cores (index-scan slicing,
    sparql.hpp:210-221): results must equal mt=1 for every suite query
    at factors beyond the index sizes."""
    from wukong_amd import queries as Q
    from tests.oracle_util import OracleCtx
    oc = OracleCtx(lubm2)
    for name, plan in Q.ALL.items():
        base = sort_rows(oc.run_query(plan, mt=1))
        for mt in (3, 64, 1000):
            got = sort_rows(oc.run_query(plan, mt=mt))
            assert got.shape == base.shape, (name, mt)
            assert np.array_equal(got, base), (name, mt)
