"""WatDiv-shaped templates: oracle vs brute (CPU) and GPU parity."""
import numpy as np
import pytest

import wukong_amd as wk
from wukong_amd import watdiv as W
from tests.oracle_util import OracleCtx, sort_rows


@pytest.fixture(scope="module")
def wd():
    return wk.watdiv_gen(2000, seed=42)


@pytest.fixture(scope="module")
def wd_oracle(wd):
    return OracleCtx(wd)


def test_watdiv_gen_shape(wd):
    assert wd.shape[0] > 2000 * 40
    assert wd[:, 1].max() < 15


@pytest.mark.parametrize("name", list(W.ALL))
def test_watdiv_oracle_vs_brute(name, wd_oracle):
    plan = W.ALL[name]
    a = sort_rows(wd_oracle.run_query(plan))
    b = sort_rows(wd_oracle.brute_query(plan))
    assert a.shape == b.shape, (name, a.shape, b.shape)
    assert np.array_equal(a, b)
    assert a.shape[0] > 0, name  # templates must be non-trivial


@pytest.mark.gpu
def test_watdiv_gpu_parity(wd, wd_oracle):
    store = wk.Store(wd)
    eng = wk.Engine(store, device=0)
    for name, plan in W.ALL.items():
        got = eng.run_query(plan)
        want = wd_oracle.run_query(plan)
        assert got.shape == want.shape, (name, got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), name
