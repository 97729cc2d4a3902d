"""Distributed UNION coverage (world_size=2, gloo, CPU): branch
orchestration in the dist driver — each branch inherits the main-BGP
table and runs through the full step machinery (exchanges, remote
reads, const-end filters), outputs concatenated, final ops once after
the rank merge — must equal the single-partition oracle's
ok_run_query_ex (execute_sparql_query, sparql.hpp:1564-1601; merge
rmap.hpp:57-87).  OPTIONAL stays single-GPU and must refuse loudly."""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp


def _union_plans(Q, wk):
    X, Y, Z = -1, -2, -3
    # branches local to the inherited var (no exchange inside branches)
    basic = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 2, [X, Y],
                    unions=[[(X, Q.MEMBEROF, wk.DIR_OUT, Y)],
                            [(X, Q.UGDEGREE, wk.DIR_OUT, Y)]])
    # branches starting on a NON-local var: the fork-join exchange (or
    # sub-threshold remote read) happens INSIDE each branch
    exch = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, wk.DIR_IN, X),
                    (X, Q.MEMBEROF, wk.DIR_OUT, Y)], 3, [X, Z],
                   unions=[[(Y, Q.SUBORG, wk.DIR_OUT, Z)],
                           [(Y, Q.WORKSFOR, wk.DIR_IN, Z)]])
    # final ops after the union merge (per-rank DISTINCT/LIMIT would be
    # wrong for the same reasons as plain plans)
    dlo = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 2, [X, Y],
                  distinct=True, limit=7, offset=2,
                  unions=[[(X, Q.MEMBEROF, wk.DIR_OUT, Y)],
                          [(X, Q.UGDEGREE, wk.DIR_OUT, Y)]])
    # one branch empties (departments are not grad students): the
    # concat must survive zero-row parts
    empty = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 2, [X, Y],
                    unions=[[(X, Q.MEMBEROF, wk.DIR_OUT, Y),
                             (Y, Q.TYPE_ID, wk.DIR_OUT, Q.GRADSTUDENT)],
                            [(X, Q.UGDEGREE, wk.DIR_OUT, Y)]])
    # predicate-variable branches (vu inside a union: both bind the
    # same pred/obj vars, so the branch layouts agree)
    vu = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, wk.DIR_IN, X)], 3, [X, Y, Z],
                 unions=[[(X, Y, wk.DIR_OUT, Z)],
                         [(X, Y, wk.DIR_IN, Z)]])
    return {"basic": basic, "exch": exch, "dlo": dlo, "empty": empty,
            "vu": vu}


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
        ctxs = [OracleCtx(wk.lubm_gen(2, seed=42, sid=r, nsrv=world),
                          sid=r, nsrv=world) for r in range(world)]
        ctx = ctxs[rank]
        out = {}
        for thr_name, thr, peers in (("x", 0, None), ("r", 10**9, ctxs)):
            for name, plan in _union_plans(Q, wk).items():
                ex = OracleExecutor(ctx, plan, peers=peers)
                dq = DistQuery(ex, plan, rank, world, threshold=thr)
                dq.run()
                out[f"{thr_name}:{name}"] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_dist_union_two_ranks_equal_single():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29878
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
    for name, plan in _union_plans(Q, wk).items():
        want = sort_rows(full.run_query(plan))
        assert len(want) > 0 or name == "never", name  # plans are non-trivial
        for mode in ("x", "r"):
            g = got[f"{mode}:{name}"]
            assert g.shape == want.shape, (mode, name, g.shape, want.shape)
            assert np.array_equal(g, want), (mode, name)


def _worker4(rank, world, port, results):
    """World-4: union + versatile plans through the exchange path (the
    4/8-GPU bench shape) — owner math must hold beyond 2 ranks."""
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
        full = OracleCtx(wk.lubm_gen(2, seed=42))
        plans = dict(_union_plans(Q, wk))
        plans.update({f"v_{n}": p
                      for n, p in Q.versatile_plans(full).items()})
        from tests.test_dist_optional import _opt_plans
        plans.update({f"o_{n}": p for n, p in _opt_plans(Q, wk).items()})
        ctx = OracleCtx(wk.lubm_gen(2, seed=42, sid=rank, nsrv=world),
                        sid=rank, nsrv=world)
        out = {}
        for name, plan in plans.items():
            ex = OracleExecutor(ctx, plan)
            dq = DistQuery(ex, plan, rank, world)
            dq.run()
            out[name] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_dist_union_versatile_four_ranks_equal_single():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29880
    procs = [ctx.Process(target=_worker4, args=(r, 4, port, results))
             for r in range(4)]
    for p in procs:
        p.start()
    got = results.get(timeout=540)
    for p in procs:
        p.join(timeout=60)

    import wukong_amd as wk
    from wukong_amd import queries as Q
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    plans = dict(_union_plans(Q, wk))
    plans.update({f"v_{n}": p for n, p in Q.versatile_plans(full).items()})
    from tests.test_dist_optional import _opt_plans
    plans.update({f"o_{n}": p for n, p in _opt_plans(Q, wk).items()})
    for name, plan in plans.items():
        want = sort_rows(full.run_query(plan))
        assert got[name].shape == want.shape, (name, got[name].shape,
                                               want.shape)
        assert np.array_equal(got[name], want), name
