"""Multi-process (world_size=2, gloo) test of the distributed driver:
2 partitioned oracle executors + all-to-all exchange must equal the
1-partition oracle.  Runs on CPU."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

QUERIES = ["q1", "q2", "q3", "q4", "q5", "q6", "q7"]


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
        triples = wk.lubm_gen(2, seed=42, sid=rank, nsrv=world)
        ctx = OracleCtx(triples, sid=rank, nsrv=world)
        out = {}
        for name in QUERIES:
            plan = Q.ALL[name]
            ex = OracleExecutor(ctx, plan)
            dq = DistQuery(ex, plan, rank, world)
            dq.run()
            merged = dq.gather_result()
            out[name] = sort_rows(merged)
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


def _worker_modifiers(rank, world, port, results):
    """Plans with DISTINCT/LIMIT/OFFSET and a mid-plan const-start
    membership filter: final ops must run ONCE after the rank merge
    (sparql.hpp:1424), and the constant's edge list must be broadcast
    from its owner rank (gstore.hpp:260-338 one-sided-read analog)."""
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
        triples = wk.lubm_gen(2, seed=42, sid=rank, nsrv=world)
        ctx = OracleCtx(triples, sid=rank, nsrv=world)
        out = {}
        for name, plan in _modifier_plans(Q, wk).items():
            ex = OracleExecutor(ctx, plan)
            dq = DistQuery(ex, plan, rank, world)
            dq.run()
            out[name] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


def _modifier_plans(Q, wk):
    """Shapes the advisor flagged: DISTINCT (cross-rank duplicates),
    DISTINCT+OFFSET+LIMIT (world*limit rows), and a const-start
    membership filter after step 0 (owner-only edge list)."""
    X, Z = -1, -2
    distinct = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, Q.DIR_IN, X),
                        (X, Q.MEMBEROF, Q.DIR_OUT, Z)],
                       nvars=2, required_vars=[Z], distinct=True)
    dlo = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, Q.DIR_IN, X),
                   (X, Q.MEMBEROF, Q.DIR_OUT, Z)],
                  nvars=2, required_vars=[Z], distinct=True,
                  limit=5, offset=3)
    cstart = wk.Plan([(Q.FULLPROF, Q.TYPE_ID, Q.DIR_IN, X),
                      (Q.DEPT0_UNIV0, Q.WORKSFOR, Q.DIR_IN, X)],
                     nvars=1, required_vars=[X])
    # final-op edge values: LIMIT 0 and OFFSET past the end must give
    # empty results once, after the rank merge
    lim0 = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, Q.DIR_IN, X),
                    (X, Q.MEMBEROF, Q.DIR_OUT, Z)],
                   nvars=2, required_vars=[Z], distinct=True, limit=0)
    offbig = wk.Plan([(Q.GRADSTUDENT, Q.TYPE_ID, Q.DIR_IN, X),
                      (X, Q.MEMBEROF, Q.DIR_OUT, Z)],
                     nvars=2, required_vars=[Z], distinct=True,
                     offset=10**6)
    return {"distinct": distinct, "distinct_lim_off": dlo,
            "const_mid": cstart, "lim0": lim0, "offbig": offbig}


def _worker_remote(rank, world, port, results):
    """Small-table remote-read path (need_fork_join threshold,
    sparql.hpp:802-814): with a huge threshold every sub-query avoids
    the exchange and probes the owner partition in place (oracle peers
    = the xGMI stand-in); results must equal both the exchange path and
    the single-partition oracle."""
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
        for thr_name, thr in (("remote", 10**9), ("mixed", 300)):
            for name in QUERIES:
                plan = Q.ALL[name]
                ex = OracleExecutor(ctx, plan, peers=ctxs)
                dq = DistQuery(ex, plan, rank, world, threshold=thr)
                dq.run()
                out[f"{thr_name}:{name}"] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_dist_remote_reads_equal_single():
    """Both sides of the rdma_threshold gate: threshold=inf forces the
    in-place remote-read path on every eligible step; threshold=300
    (the reference default) mixes remote and exchange."""
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29875
    procs = [ctx.Process(target=_worker_remote, args=(r, 2, port, results))
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
    for name in QUERIES:
        want = sort_rows(full.run_query(Q.ALL[name]))
        for mode in ("remote", "mixed"):
            g = got[f"{mode}:{name}"]
            assert g.shape == want.shape, (mode, name, g.shape, want.shape)
            assert np.array_equal(g, want), (mode, name)


@pytest.mark.timeout(600)
def test_dist_modifiers_and_const_start():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29874
    procs = [ctx.Process(target=_worker_modifiers, args=(r, 2, port, results))
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
    for name, plan in _modifier_plans(Q, wk).items():
        want = sort_rows(full.run_query(plan))
        assert got[name].shape == want.shape, (name, got[name].shape, want.shape)
        assert np.array_equal(got[name], want), name


@pytest.mark.timeout(600)
def test_dist_two_ranks_equal_single():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29871
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
    for name in QUERIES:
        want = sort_rows(full.run_query(Q.ALL[name]))
        assert got[name].shape == want.shape, (name, got[name].shape, want.shape)
        assert np.array_equal(got[name], want), name


@pytest.mark.timeout(600)
def test_dist_three_ranks_equal_single():
    """world_size=3 — odd world: vid % 3 ownership has no power-of-two
    structure to hide modulo mistakes behind."""
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29882
    procs = [ctx.Process(target=_worker, args=(r, 3, port, results))
             for r in range(3)]
    for p in procs:
        p.start()
    got = results.get(timeout=540)
    for p in procs:
        p.join(timeout=60)

    import wukong_amd as wk
    from wukong_amd import queries as Q
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    for name in QUERIES:
        want = sort_rows(full.run_query(Q.ALL[name]))
        assert got[name].shape == want.shape, (name, got[name].shape, want.shape)
        assert np.array_equal(got[name], want), name


@pytest.mark.timeout(600)
def test_dist_four_ranks_equal_single():
    """world_size=4 — the shape the driver's 4/8-GPU scaling bench uses."""
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29872
    procs = [ctx.Process(target=_worker, args=(r, 4, port, results))
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
    for name in QUERIES:
        want = sort_rows(full.run_query(Q.ALL[name]))
        assert got[name].shape == want.shape, (name, got[name].shape, want.shape)
        assert np.array_equal(got[name], want), name


@pytest.mark.timeout(900)
def test_dist_eight_ranks_equal_single():
    """world_size=8 — the driver's full-node scaling shape."""
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29873
    procs = [ctx.Process(target=_worker, args=(r, 8, port, results))
             for r in range(8)]
    for p in procs:
        p.start()
    got = results.get(timeout=840)
    for p in procs:
        p.join(timeout=60)

    import wukong_amd as wk
    from wukong_amd import queries as Q
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    for name in QUERIES:
        want = sort_rows(full.run_query(Q.ALL[name]))
        assert got[name].shape == want.shape, (name, got[name].shape, want.shape)
        assert np.array_equal(got[name], want), name
