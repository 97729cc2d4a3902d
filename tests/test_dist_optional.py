"""Distributed OPTIONAL coverage (world_size=2, gloo, CPU): the
driver's host-side restatement of the matched-flag row mechanics
(sparql.hpp:100-170,316-375,416-549 under :1603-1662) with
owner-correct probes (row exchange per step, broadcast const lists)
must equal the single-partition oracle's ok_run_query_ex."""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp


def _opt_plans(Q, wk):
    X, Y, Z = -1, -2, -3
    # left join: students with their advisor (or BLANK)
    k2u = wk.Plan([(Q.UGSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 2, [X, Y],
                  optional=[(X, Q.ADVISOR, wk.DIR_OUT, Y)])
    # + typeof filter on the optional-born column (blank + unmatch)
    k2c = wk.Plan([(Q.UGSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 2, [X, Y],
                  optional=[(X, Q.ADVISOR, wk.DIR_OUT, Y),
                            (Y, Q.TYPE_ID, wk.DIR_OUT, Q.FULLPROF)])
    # k2k inside the group: advisor works for the student's department
    k2k = wk.Plan([(Q.UGSTUDENT, Q.TYPE_ID, wk.DIR_IN, X),
                   (X, Q.MEMBEROF, wk.DIR_OUT, Z)], 3, [X, Y, Z],
                  optional=[(X, Q.ADVISOR, wk.DIR_OUT, Y),
                            (Y, Q.WORKSFOR, wk.DIR_OUT, Z)])
    # const_to_known inside the group (owner-only edge list broadcast)
    cfilter = wk.Plan([(Q.UGSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 2, [X, Y],
                      optional=[(X, Q.ADVISOR, wk.DIR_OUT, Y),
                                (Q.DEPT0_UNIV0, Q.WORKSFOR, wk.DIR_IN, Y)])
    # union THEN optional (the reference's execution order)
    uo = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 3, [X, Y, Z],
                 unions=[[(X, Q.MEMBEROF, wk.DIR_OUT, Y)],
                         [(X, Q.UGDEGREE, wk.DIR_OUT, Y)]],
                 optional=[(X, Q.ADVISOR, wk.DIR_OUT, Z)])
    # final ops over the optional result (once, after the rank merge)
    dlo = wk.Plan([(Q.UGSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 2, [Y],
                  distinct=True, limit=9, offset=4,
                  optional=[(X, Q.ADVISOR, wk.DIR_OUT, Y)])
    # 3-pattern group: chain over two group-born columns + typeof on
    # the second (unmatch must blank BOTH opt columns)
    W2 = -4
    chain = wk.Plan([(Q.UGSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 4, [X, Y, W2],
                    optional=[(X, Q.ADVISOR, wk.DIR_OUT, Y),
                              (Y, Q.WORKSFOR, wk.DIR_OUT, W2),
                              (W2, Q.TYPE_ID, wk.DIR_OUT, Q.DEPARTMENT)])
    return {"k2u": k2u, "k2c": k2c, "k2k": k2k, "cfilter": cfilter,
            "uo": uo, "dlo": dlo, "chain": chain}


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
        ctx = OracleCtx(wk.lubm_gen(2, seed=42, sid=rank, nsrv=world),
                        sid=rank, nsrv=world)
        out = {}
        for name, plan in _opt_plans(Q, wk).items():
            ex = OracleExecutor(ctx, plan)
            dq = DistQuery(ex, plan, rank, world)
            dq.run()
            out[name] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_dist_optional_two_ranks_equal_single():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29881
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
    for name, plan in _opt_plans(Q, wk).items():
        want = sort_rows(full.run_query(plan))
        assert len(want) > 0, name
        g = got[name]
        assert g.shape == want.shape, (name, g.shape, want.shape)
        assert np.array_equal(g, want), name


def test_dist_optional_vu_refuses():
    """Predicate variables inside an OPTIONAL group don't fit the
    row-owner exchange (and no reference plan uses them there) — the
    driver refuses the shape."""
    import wukong_amd as wk
    from wukong_amd import queries as Q
    from wukong_amd.dist import DistQuery

    class _NullEx:
        def step(self):
            return 0

        def rows(self):
            return 0

        def table(self):
            return np.empty((0, 1), dtype=np.uint32)

    plan = wk.Plan([(Q.UGSTUDENT, Q.TYPE_ID, wk.DIR_IN, -1)], 3, [-1],
                   optional=[(-1, -2, wk.DIR_OUT, -3)])
    dq = DistQuery(_NullEx(), plan, 0, 1)
    with pytest.raises(ValueError):
        dq.run()
