"""Randomized plan fuzzing: generated BGPs over LUBM-2 must agree
across the oracle, the independent brute evaluator, and (on GPU) the
HIP engine — the operator-level parity net beyond the fixed Q1-Q12."""
import random

import numpy as np
import pytest

import wukong_amd as wk
from wukong_amd import Plan, queries as Q
from tests.oracle_util import sort_rows

PREDS = [Q.SUBORG, Q.UGDEGREE, Q.MEMBEROF, Q.WORKSFOR, Q.TEACHEROF,
         Q.ADVISOR, Q.TAKESCOURSE, Q.HEADOF, Q.PUBAUTHOR]
TYPES = [Q.UNIVERSITY, Q.DEPARTMENT, Q.FULLPROF, Q.ASSTPROF, Q.UGSTUDENT,
         Q.GRADSTUDENT, Q.COURSE, Q.GRADCOURSE, Q.RESEARCHGROUP,
         Q.PUBLICATION]


def random_plan(rng, store):
    """A random valid plan: index/const start, then known-start
    expansions/filters, occasionally a predicate variable."""
    nv = rng.randint(2, 5)
    vars_ = [-(i + 1) for i in range(nv)]
    pats = []
    bound = []
    # start
    if rng.random() < 0.5:
        t = rng.choice(TYPES)
        pats.append((t, Q.TYPE_ID, wk.DIR_IN, vars_[0]))
    else:
        pool = store.get_index(rng.choice(TYPES), wk.DIR_IN)
        const = int(rng.choice(list(pool)))
        pats.append((const, rng.choice(PREDS), rng.choice([0, 1]), vars_[0]))
    bound.append(vars_[0])
    free = [v for v in vars_ if v not in bound]
    for _ in range(rng.randint(1, 4)):
        s = rng.choice(bound)
        r = rng.random()
        if r < 0.18 and free:   # k2u with a predicate variable
            pvar = free.pop(0)
            if free:
                ovar = free.pop(0)
                pats.append((s, pvar, rng.choice([0, 1]), ovar))
                bound += [pvar, ovar]
            else:
                pats.append((s, pvar, rng.choice([0, 1]),
                             int(rng.choice(TYPES)) if False else
                             (1 << 17)))
                bound.append(pvar)
        elif r < 0.5 and free:  # k2u
            o = free.pop(0)
            pats.append((s, rng.choice(PREDS), rng.choice([0, 1]), o))
            bound.append(o)
        elif r < 0.75:          # typeof / const filter
            pats.append((s, Q.TYPE_ID, wk.DIR_OUT, rng.choice(TYPES)))
        else:                   # k2k or k2c
            o = rng.choice(bound)
            if o != s and rng.random() < 0.6:
                pats.append((s, rng.choice(PREDS), rng.choice([0, 1]), o))
            else:
                pats.append((s, rng.choice(PREDS), rng.choice([0, 1]),
                             1 << 17))
    req = [v for v in bound]
    return Plan(pats, nvars=nv, required_vars=req)


@pytest.fixture(scope="module")
def fuzz_store(lubm2):
    return wk.Store(lubm2)


def test_fuzz_oracle_vs_brute(lubm2, fuzz_store):
    from tests.oracle_util import OracleCtx
    oc = OracleCtx(lubm2)
    rng = random.Random(1234)
    ran = 0
    for trial in range(60):
        plan = random_plan(rng, fuzz_store)
        try:
            a = oc.run_query(plan)
        except RuntimeError:
            continue  # oracle rejects the shape: fine, skip
        b = oc.brute_query(plan)
        assert a.shape == b.shape, (trial, plan.patterns, a.shape, b.shape)
        assert np.array_equal(sort_rows(a), sort_rows(b)), \
            (trial, plan.patterns)
        ran += 1
    assert ran > 30


@pytest.mark.gpu
def test_fuzz_gpu_vs_oracle(lubm2, fuzz_store):
    from tests.oracle_util import OracleCtx
    oc = OracleCtx(lubm2)
    eng = wk.Engine(fuzz_store, device=0)
    rng = random.Random(99)
    ran = 0
    for trial in range(60):
        plan = random_plan(rng, fuzz_store)
        try:
            want = oc.run_query(plan)
        except RuntimeError:
            continue
        try:
            got = eng.run_query(plan)
        except RuntimeError as ex:
            raise AssertionError((trial, plan.patterns, str(ex)))
        assert got.shape == want.shape, (trial, plan.patterns,
                                         got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), \
            (trial, plan.patterns)
        ran += 1
    assert ran > 30


def test_fuzz_watdiv_oracle_vs_brute():
    """Same fuzz net over the WatDiv schema (different shape: hubs,
    deep chains, genre fan-outs)."""
    from tests.oracle_util import OracleCtx
    from wukong_amd import watdiv as W
    tri = wk.watdiv_gen(2000, seed=7)
    store = wk.Store(tri)
    oc = OracleCtx(tri)
    preds = [W.HASGENRE, W.OFFER_PRODUCT, W.RETAILER, W.REVIEW_PRODUCT,
             W.REVIEWER, W.PURCHASED, W.FRIEND]
    types = [W.T_PRODUCT, W.T_OFFER, W.T_REVIEW, W.T_USER, W.T_GENRE,
             W.T_RETAILER]
    rng = random.Random(5)
    ran = 0
    for trial in range(40):
        nv = rng.randint(2, 4)
        vars_ = [-(i + 1) for i in range(nv)]
        t = rng.choice(types)
        pats = [(t, Q.TYPE_ID, wk.DIR_IN, vars_[0])]
        bound = [vars_[0]]
        free = vars_[1:]
        for _ in range(rng.randint(1, 3)):
            s = rng.choice(bound)
            r = rng.random()
            if r < 0.15 and len(free) >= 2:
                # predicate variable over the WatDiv vp lists
                pvar, ovar = free.pop(0), free.pop(0)
                pats.append((s, pvar, rng.choice([0, 1]), ovar))
                bound += [pvar, ovar]
            elif r < 0.55 and free:
                o = free.pop(0)
                pats.append((s, rng.choice(preds), rng.choice([0, 1]), o))
                bound.append(o)
            else:
                pats.append((s, Q.TYPE_ID, wk.DIR_OUT, rng.choice(types)))
        plan = Plan(pats, nvars=nv, required_vars=bound)
        try:
            a = oc.run_query(plan)
        except RuntimeError:
            continue
        b = oc.brute_query(plan)
        assert a.shape == b.shape, (trial, pats, a.shape, b.shape)
        assert np.array_equal(sort_rows(a), sort_rows(b)), (trial, pats)
        ran += 1
    assert ran > 20
