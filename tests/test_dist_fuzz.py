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
