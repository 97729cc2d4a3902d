"""VERSATILE (predicate-variable) coverage — LUBM Q8-Q12
(sparql.hpp:556-744; vp lists static_gstore.hpp:282-374).

CPU: oracle vs the independent brute evaluator; native store's dense vp
CSR vs the oracle's hash-restatement lists.  GPU (-m gpu): HIP engine
vs oracle, set-equal."""
import numpy as np
import pytest

import wukong_amd as wk
from wukong_amd import queries as Q
from tests.oracle_util import OracleCtx, sort_rows


@pytest.fixture(scope="module")
def vplans4(oracle4):
    return Q.versatile_plans(oracle4)


def test_oracle_vs_brute_q8_q12(oracle4, vplans4):
    for name, plan in vplans4.items():
        got = oracle4.run_query(plan)
        want = oracle4.brute_query(plan)
        assert got.shape == want.shape, (name, got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), name
        if name not in ("q12",):  # the p-var queries have rows on LUBM-4
            assert len(got) > 0, name


def test_store_vp_lists_match_oracle(lubm4, oracle4):
    """Native dense vp CSR == oracle [vid|PREDICATE_ID|dir] lists for
    every vid present (both directions, ascending pid)."""
    store = wk.Store(lubm4)
    vids = np.unique(lubm4[:, [0, 2]])
    vids = vids[vids >= (1 << 17)]
    checked = 0
    for d in (wk.DIR_OUT, wk.DIR_IN):
        for v in vids:
            want = np.asarray(oracle4.get_triples(int(v), 0, d))
            got = np.asarray(store.get_triples(int(v), 0, d))
            assert np.array_equal(got, want), (int(v), d, got, want)
            checked += 1
    assert checked > 100


def test_oracle_vp_semantics(lubm4, oracle4):
    """OUT lists include rdf:type; IN lists never do (type triples are
    index-only on the IN side — static_gstore.hpp:336-339)."""
    store = wk.Store(lubm4)
    some_typed = int(np.asarray(store.get_index(Q.UNIVERSITY, wk.DIR_IN))[0])
    out_preds = np.asarray(store.get_triples(some_typed, 0, wk.DIR_OUT))
    assert Q.TYPE_ID in out_preds
    vids = np.unique(lubm4[:, 2])
    vids = vids[vids >= (1 << 17)]
    for v in vids[:50]:
        in_preds = np.asarray(store.get_triples(int(v), 0, wk.DIR_IN))
        assert Q.TYPE_ID not in in_preds


@pytest.mark.gpu
def test_gpu_versatile_parity(store4, oracle4):
    vplans = Q.versatile_plans(store4)
    eng = wk.Engine(store4, device=0)
    for name, plan in vplans.items():
        got = eng.run_query(plan)
        want = oracle4.run_query(plan)
        assert got.shape == want.shape, (name, got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), name


@pytest.mark.gpu
def test_gpu_versatile_counts_lubm40():
    """Bigger store (LUBM-40): engine vs oracle row counts + checksum."""
    triples = wk.lubm_gen(40, seed=42)
    store = wk.Store(triples)
    oc = OracleCtx(triples)
    vplans = Q.versatile_plans(store)
    eng = wk.Engine(store, device=0)
    for name, plan in vplans.items():
        got = eng.run_query(plan)
        want = oc.run_query(plan)
        assert got.shape == want.shape, (name, got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), name
