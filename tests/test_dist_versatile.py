"""Distributed VERSATILE (predicate-variable) coverage: LUBM Q8-Q12
across 2 ranks (gloo, CPU) must equal the single-partition oracle.

The vp list of a vertex lives on its owner rank (the partition that
holds its pso/pos runs, base_loader.hpp:284), so: const_unknown_* at
step 0 yields rows only on the owner (concat across ranks = the single
answer); known-start vu steps exchange on the start var first, then
read the owner-local vp CSR (sparql.hpp:556-744 semantics)."""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

VQUERIES = ["q8", "q9", "q10", "q11", "q12"]


def _worker(rank, world, port, results):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch.distributed as dist
    import wukong_amd as wk
    from wukong_amd import queries as Q
    from wukong_amd.dist import DistQuery
    from tests.oracle_util import OracleCtx, OracleExecutor, sort_rows

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # plans are derived positionally from the FULL store (constants
        # like "dept 3 of univ 0"), identical in every process
        full = OracleCtx(wk.lubm_gen(2, seed=42))
        plans = {n: p for n, p in
                 zip(["q8", "q9", "q10", "q11", "q12"],
                     [Q.versatile_plans(full)[k]
                      for k in ["q8", "q9", "q10", "q11", "q12"]])}
        ctx = OracleCtx(wk.lubm_gen(2, seed=42, sid=rank, nsrv=world),
                        sid=rank, nsrv=world)
        out = {}
        for name in VQUERIES:
            plan = plans[name]
            ex = OracleExecutor(ctx, plan)
            dq = DistQuery(ex, plan, rank, world)
            dq.run()
            out[name] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_dist_versatile_two_ranks_equal_single():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29877
    procs = [ctx.Process(target=_worker, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
        p.start()
    got = results.get(timeout=540)
    for p in procs:
        p.join(timeout=60)

    import wukong_amd as wk
    from wukong_amd import queries as Q
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    plans = Q.versatile_plans(full)
    nonempty = 0
    for name in VQUERIES:
        want = sort_rows(full.run_query(plans[name]))
        g = got[name]
        assert g.shape == want.shape, (name, g.shape, want.shape)
        assert np.array_equal(g, want), name
        nonempty += bool(len(want))
    assert nonempty >= 3  # the suite is non-trivial on LUBM-2


def test_dist_mid_plan_const_vu_refuses():
    """const_unknown_* must be the first pattern (sparql.hpp:719; the
    engine rejects it mid-plan with WK_ERR_PLAN) — the dist driver
    refuses the shape up front rather than broadcasting a nonsense
    membership filter."""
    import wukong_amd as wk
    from wukong_amd import queries as Q
    from wukong_amd.dist import DistQuery

    class _NullEx:
        def step(self):
            return 0

    plan = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, wk.DIR_IN, -1),
                    (Q.DEPT0_UNIV0, -2, wk.DIR_IN, -3)],
                   3, [-1, -2, -3])
    dq = DistQuery(_NullEx(), plan, 0, 1)
    with pytest.raises(ValueError):
        dq.run()
