"""Entities with MULTIPLE rdf:type values: the dense type index must
mark them 0xFFFF and every type filter must fall back to the probe path
(and the light2 single-kernel path must be disabled store-wide)."""
import numpy as np
import pytest

import wukong_amd as wk
from wukong_amd import Plan
from tests.oracle_util import OracleCtx, sort_rows

TYPE = 1
P = 5           # a predicate
T1, T2 = 20, 21  # type ids
B = 1 << 17


def _triples():
    # A has TWO types; D has none; edges A->B->C, D->A
    A, Bv, C, D, E = B + 1, B + 2, B + 3, B + 4, B + 5
    t = [
        (A, TYPE, T1), (A, TYPE, T2),
        (Bv, TYPE, T1),
        (C, TYPE, T2),
        (E, TYPE, T1),
        (A, P, Bv), (Bv, P, C), (D, P, A), (A, P, E), (Bv, P, E),
    ]
    return np.array(t, dtype=np.uint32)


PLANS = {
    # all T1 members -> follow P -> keep targets of type T1
    "p1": Plan([(T1, TYPE, 0, -1), (-1, P, 1, -2), (-2, TYPE, 1, T1)],
               nvars=2, required_vars=[-1, -2]),
    # filter on the multi-typed entity itself: A must pass BOTH T1 and T2
    "p2": Plan([(T1, TYPE, 0, -1), (-1, TYPE, 1, T2)],
               nvars=1, required_vars=[-1]),
    # light2-shaped plan (const start + type filter) on a multi-type store
    "p3": Plan([((1 << 17) + 2, P, 0, -1), (-1, TYPE, 1, T1)],
               nvars=1, required_vars=[-1]),
}


def test_multitype_oracle_vs_brute():
    t = _triples()
    ora = OracleCtx(t)
    for name, plan in PLANS.items():
        a = sort_rows(ora.run_query(plan))
        b = sort_rows(ora.brute_query(plan))
        assert np.array_equal(a, b), name


@pytest.mark.gpu
def test_multitype_gpu_parity():
    t = _triples()
    store = wk.Store(t)
    eng = wk.Engine(store, device=0)
    ora = OracleCtx(t)
    for name, plan in PLANS.items():
        got = eng.run_query(plan)
        want = ora.run_query(plan)
        assert got.shape == want.shape, (name, got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), name
        # submit path too (light2 must be inhibited by type_multi)
        eng.submit(plan)
        got2 = eng.fetch_result()
        assert np.array_equal(sort_rows(got2), sort_rows(want)), name
