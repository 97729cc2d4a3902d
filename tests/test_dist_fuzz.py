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
from wukong_amd import queries as Q
from tests.test_fuzz_plans import random_plan
from tests.oracle_util import OracleCtx, sort_rows

N_PLANS = 25


def _gen_plans():
    """Deterministic plan list (no predicate variables: the per-pattern
    distributed driver covers the BGP surface)."""
    store = wk.Store(wk.lubm_gen(2, seed=42))
    rng = random.Random(4321)
    plans = []
    while len(plans) < N_PLANS:
        p = random_plan(rng, store)
        if all(pp[1] >= 1 for pp in p.patterns):
            plans.append(p)
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
