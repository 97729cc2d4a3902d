"""Distributed fuzz: random BGPs through the FULL dist driver (2-rank
gloo, oracle executors) on both sides of the rdma_threshold gate must
equal the single-partition oracle — the operator-level parity net for
the exchange, remote-read (peer) and broadcast-filter paths beyond the
fixed Q1-Q7 suites."""
import os
import random

import numpy as np
import pytest
import torch.multiprocessing as mp

import wukong_amd as wk
from wukong_amd import Plan, queries as Q
from tests.test_fuzz_plans import random_plan, PREDS, TYPES
from tests.oracle_util import OracleCtx, sort_rows

N_PLANS = 25
N_VU = 8      # plans containing a predicate variable (distributed vu)
N_UNION = 8   # plans with random UNION branches
N_OPT = 8     # plans with a random OPTIONAL group


def _gen_plans():
    """Deterministic plan list: plain BGPs, BGPs with a predicate
    variable (the distributed vu path), and BGPs with two random
    layout-compatible UNION branches (expansion branches share the
    start/new var so every branch appends the same column; filter
    branches append none)."""
    store = wk.Store(wk.lubm_gen(2, seed=42))
    rng = random.Random(4321)
    plans = []
    while len(plans) < N_PLANS:
        p = random_plan(rng, store)
        if all(pp[1] >= 1 for pp in p.patterns):
            plans.append(p)
    rng2 = random.Random(8765)
    while len(plans) < N_PLANS + N_VU:
        p = random_plan(rng2, store)
        # vu steps are always known-start in random_plan (mid-plan
        # const vu is refused by driver and engine alike)
        if any(pp[1] < 0 for pp in p.patterns):
            plans.append(p)
    rng3 = random.Random(1357)
    n_uni = 0
    while n_uni < N_UNION:
        base = random_plan(rng3, store)
        if not all(pp[1] >= 1 for pp in base.patterns):
            continue
        bound = list(base.required_vars)
        nv = base.nvars
        if rng3.random() < 0.5:
            s = rng3.choice(bound)
            ovar = -(nv + 1)
            branches = [[(s, rng3.choice(PREDS), rng3.choice([0, 1]), ovar)]
                        for _ in range(2)]
            plan = Plan(list(base.patterns), nv + 1, bound + [ovar],
                        unions=branches)
        else:
            branches = [[(rng3.choice(bound), Q.TYPE_ID, wk.DIR_OUT,
                          rng3.choice(TYPES))] for _ in range(2)]
            plan = Plan(list(base.patterns), nv, bound, unions=branches)
        plans.append(plan)
        n_uni += 1
    rng4 = random.Random(2468)
    n_opt = 0
    while n_opt < N_OPT:
        base = random_plan(rng4, store)
        if not all(pp[1] >= 1 for pp in base.patterns):
            continue
        bound = list(base.required_vars)
        nv = base.nvars
        s = rng4.choice(bound)
        ovar = -(nv + 1)
        group = [(s, rng4.choice(PREDS), rng4.choice([0, 1]), ovar)]
        r = rng4.random()
        if r < 0.4:    # typeof filter on the optional-born column
            group.append((ovar, Q.TYPE_ID, wk.DIR_OUT, rng4.choice(TYPES)))
        elif r < 0.6:  # k2k back onto a base column
            other = rng4.choice(bound)
            group.append((ovar, rng4.choice(PREDS), rng4.choice([0, 1]),
                          other))
        plan = Plan(list(base.patterns), nv + 1, bound + [ovar],
                    optional=group)
        plans.append(plan)
        n_opt += 1
    return plans


def _worker(rank, world, port, results):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch.distributed as dist
    from wukong_amd.dist import DistQuery
    from tests.oracle_util import OracleExecutor

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        ctxs = [OracleCtx(wk.lubm_gen(2, seed=42, sid=r, nsrv=world),
                          sid=r, nsrv=world) for r in range(world)]
        out = {}
        for j, plan in enumerate(_gen_plans()):
            for mode, thr in (("x", 0), ("r", 10**9)):
                ex = OracleExecutor(ctxs[rank], plan, peers=ctxs)
                dq = DistQuery(ex, plan, rank, world, threshold=thr)
                dq.run()
                out[f"{mode}:{j}"] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_dist_fuzz_two_ranks():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29876, results))
             for r in range(2)]
    for p in procs:
        p.start()
    got = results.get(timeout=840)
    for p in procs:
        p.join(timeout=60)

    full = OracleCtx(wk.lubm_gen(2, seed=42))
    for j, plan in enumerate(_gen_plans()):
        want = sort_rows(full.run_query(plan))
        for mode in ("x", "r"):
            g = got[f"{mode}:{j}"]
            assert g.shape == want.shape, (mode, j, g.shape, want.shape,
                                           plan.patterns)
            assert np.array_equal(g, want), (mode, j, plan.patterns)


def _gen_watdiv_plans():
    """WatDiv-schema random WALKS (hubs, chains, genre fan-outs) —
    schema-aware so most plans are non-empty: each hop follows an edge
    the generator actually emits (watdiv_gen.cpp), tracking the bound
    var's type."""
    from wukong_amd import watdiv as W
    # type -> [(pred, dir, result_type)] per the generator's schema
    EDGES = {
        W.T_PRODUCT: [(W.HASGENRE, wk.DIR_OUT, W.T_GENRE),
                      (W.OFFER_PRODUCT, wk.DIR_IN, W.T_OFFER),
                      (W.REVIEW_PRODUCT, wk.DIR_IN, W.T_REVIEW),
                      (W.PURCHASED, wk.DIR_IN, W.T_USER)],
        W.T_OFFER: [(W.OFFER_PRODUCT, wk.DIR_OUT, W.T_PRODUCT),
                    (W.RETAILER, wk.DIR_OUT, W.T_RETAILER)],
        W.T_REVIEW: [(W.REVIEW_PRODUCT, wk.DIR_OUT, W.T_PRODUCT),
                     (W.REVIEWER, wk.DIR_OUT, W.T_USER)],
        W.T_USER: [(W.PURCHASED, wk.DIR_OUT, W.T_PRODUCT),
                   (W.FRIEND, wk.DIR_OUT, W.T_USER),
                   (W.FRIEND, wk.DIR_IN, W.T_USER),
                   (W.REVIEWER, wk.DIR_IN, W.T_REVIEW)],
        W.T_GENRE: [(W.HASGENRE, wk.DIR_IN, W.T_PRODUCT)],
        W.T_RETAILER: [(W.RETAILER, wk.DIR_IN, W.T_OFFER)],
    }
    rng = random.Random(97)
    plans = []
    for _ in range(15):
        nv = rng.randint(2, 4)
        vars_ = [-(i + 1) for i in range(nv)]
        t = rng.choice(list(EDGES))
        pats = [(t, Q.TYPE_ID, wk.DIR_IN, vars_[0])]
        bound = {vars_[0]: t}
        free = vars_[1:]
        for _ in range(rng.randint(1, 3)):
            s = rng.choice(list(bound))
            pred, d, rt = rng.choice(EDGES[bound[s]])
            if rng.random() < 0.75 and free:
                o = free.pop(0)
                pats.append((s, pred, d, o))
                bound[o] = rt
            else:  # matching typeof filter (kept) or a wrong one (empties)
                wrong = rng.random() < 0.15
                ft = rng.choice(list(EDGES)) if wrong else bound[s]
                pats.append((s, Q.TYPE_ID, wk.DIR_OUT, ft))
        plans.append(Plan(pats, nvars=nv, required_vars=list(bound)))
    return plans


def _worker_watdiv(rank, world, port, results):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch.distributed as dist
    from wukong_amd.dist import DistQuery
    from tests.oracle_util import OracleExecutor

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        ctxs = [OracleCtx(wk.watdiv_gen(2000, seed=7, sid=r, nsrv=world),
                          sid=r, nsrv=world) for r in range(world)]
        out = {}
        for j, plan in enumerate(_gen_watdiv_plans()):
            for mode, thr in (("x", 0), ("r", 10**9)):
                ex = OracleExecutor(ctxs[rank], plan, peers=ctxs)
                dq = DistQuery(ex, plan, rank, world, threshold=thr)
                dq.run()
                out[f"{mode}:{j}"] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_dist_fuzz_watdiv_two_ranks():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker_watdiv, args=(r, 2, 29879, results))
             for r in range(2)]
    for p in procs:
        p.start()
    got = results.get(timeout=840)
    for p in procs:
        p.join(timeout=60)

    full = OracleCtx(wk.watdiv_gen(2000, seed=7))
    nonempty = 0
    for j, plan in enumerate(_gen_watdiv_plans()):
        want = sort_rows(full.run_query(plan))
        nonempty += bool(len(want))
        for mode in ("x", "r"):
            g = got[f"{mode}:{j}"]
            assert g.shape == want.shape, (mode, j, g.shape, want.shape,
                                           plan.patterns)
            assert np.array_equal(g, want), (mode, j, plan.patterns)
    assert nonempty >= 8


def _gen_planned(seed=808, n=10):
    """Shuffled random BGPs recovered by the planner — the planner's
    orders (filter-first, reoriented edges) through the dist driver."""
    from wukong_amd import planner
    store = wk.Store(wk.lubm_gen(2, seed=42))
    rng = random.Random(seed)
    plans = []
    while len(plans) < n:
        base = random_plan(rng, store)
        shuffled = list(base.patterns)
        rng.shuffle(shuffled)
        try:
            planned = planner.plan_patterns(store, shuffled, base.nvars,
                                            base.required_vars)
        except planner.PlannerError:
            continue
        plans.append((base, planned))
    return plans


def _worker_planned(rank, world, port, results):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch.distributed as dist
    from wukong_amd.dist import DistQuery
    from tests.oracle_util import OracleExecutor

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        ctxs = [OracleCtx(wk.lubm_gen(2, seed=42, sid=r, nsrv=world),
                          sid=r, nsrv=world) for r in range(world)]
        out = {}
        for j, (base, planned) in enumerate(_gen_planned()):
            for mode, thr in (("x", 0), ("m", 300)):
                ex = OracleExecutor(ctxs[rank], planned, peers=ctxs)
                dq = DistQuery(ex, planned, rank, world, threshold=thr)
                dq.run()
                out[f"{mode}:{j}"] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_dist_planned_plans_equal_textual():
    """Planner output (from shuffled inputs) through the 2-rank driver
    must equal the single oracle on the TEXTUAL order — crossing the
    planner's reorderings with the exchange/remote machinery."""
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker_planned, args=(r, 2, 29884, results))
             for r in range(2)]
    for p in procs:
        p.start()
    got = results.get(timeout=840)
    for p in procs:
        p.join(timeout=60)

    full = OracleCtx(wk.lubm_gen(2, seed=42))
    for j, (base, planned) in enumerate(_gen_planned()):
        want = sort_rows(full.run_query(base))
        for mode in ("x", "m"):
            g = got[f"{mode}:{j}"]
            assert g.shape == want.shape, (mode, j, base.patterns,
                                           planned.patterns)
            assert np.array_equal(g, want), (mode, j, planned.patterns)
