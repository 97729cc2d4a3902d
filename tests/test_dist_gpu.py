"""GPU executor under the distributed driver: 2 ranks over gloo, both on
cuda:0 (exchange via host).  Validates partitioned GPU stores + the
step/load/split plumbing end-to-end against the 1-partition oracle.
(The RCCL all-to-allv leg of the same driver runs in the 8-GPU bench.)"""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

QUERIES = ["q1", "q2", "q3", "q5", "q7"]


def _worker(rank, world, port, results):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch.distributed as dist
    import wukong_amd as wk
    from wukong_amd import queries as Q
    from wukong_amd.dist import DistQuery, GpuExecutor
    from tests.oracle_util import sort_rows

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        triples = wk.lubm_gen(2, seed=42, sid=rank, nsrv=world)
        store = wk.Store(triples, sid=rank, nsrv=world)
        eng = wk.Engine(store, device=0)
        out = {}
        plans = dict((n, Q.ALL[n]) for n in QUERIES)
        # + the advisor-flagged shapes: DISTINCT / LIMIT+OFFSET final
        # ops once after the merge, and a mid-plan const-start filter
        # through the device wk_engine_execute_filter_list path
        from tests.test_dist_gloo import _modifier_plans
        plans.update(_modifier_plans(Q, wk))
        # + UNION branch orchestration through GpuExecutor.rebind
        # (begin_query + load_rbuf continuation on device)
        from tests.test_dist_union import _union_plans
        plans.update({f"u_{n}": p for n, p in _union_plans(Q, wk).items()})
        # + VERSATILE (k_vu over the partitioned vp CSR; owner-local
        # const starts, exchange-then-local known starts)
        from tests.oracle_util import OracleCtx
        vfull = OracleCtx(wk.lubm_gen(2, seed=42))
        plans.update({f"v_{n}": p
                      for n, p in Q.versatile_plans(vfull).items()})
        # + OPTIONAL groups (host-side matched-flag restatement over
        # the device executor's table/get_triples)
        from tests.test_dist_optional import _opt_plans
        plans.update({f"o_{n}": p for n, p in _opt_plans(Q, wk).items()})
        for name, plan in plans.items():
            ex = GpuExecutor(eng, plan)
            dq = DistQuery(ex, plan, rank, world)
            dq.run()
            merged = dq.gather_result()
            out[name] = sort_rows(merged)
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


def _worker_peer(rank, world, port, results):
    """xGMI peer-probe path on hardware: both ranks on cuda:0 (the
    1-GPU box), stores opened across processes via HIP IPC (dmabuf
    mode, HSA_ENABLE_IPC_MODE_LEGACY=0); gloo carries only the control
    plane.  threshold=inf forces every eligible step through
    k_peer_step — the reference's in-place one-sided read."""
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch.distributed as dist
    import wukong_amd as wk
    from wukong_amd import queries as Q
    from wukong_amd.dist import DistQuery, GpuExecutor, init_peers
    from tests.oracle_util import sort_rows

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        triples = wk.lubm_gen(2, seed=42, sid=rank, nsrv=world)
        store = wk.Store(triples, sid=rank, nsrv=world)
        gstore = wk.GpuStore(store, device=0)
        eng = wk.Engine(gstore, device=0)
        ok = init_peers(gstore, store)
        out = {"_peers": ok}
        if ok:
            for name in QUERIES:
                plan = Q.ALL[name]
                ex = GpuExecutor(eng, plan)
                dq = DistQuery(ex, plan, rank, world, threshold=10**9)
                dq.run()
                out[name] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_gpu_peer_probe_two_ranks_equal_oracle():
    """k_peer_step (IPC-mapped stores) vs the 1-partition oracle."""
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker_peer, args=(r, 2, 29913, results))
             for r in range(2)]
    for p in procs:
        p.start()
    got = results.get(timeout=840)
    for p in procs:
        p.join(timeout=60)
    assert got["_peers"], "HIP IPC peer import failed on this box"

    import wukong_amd as wk
    from wukong_amd import queries as Q
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    for name in QUERIES:
        want = sort_rows(full.run_query(Q.ALL[name]))
        assert got[name].shape == want.shape, (name, got[name].shape, want.shape)
        assert np.array_equal(got[name], want), name


@pytest.mark.timeout(900)
def test_gpu_dist_two_ranks_equal_oracle():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29911, results))
             for r in range(2)]
    for p in procs:
        p.start()
    got = results.get(timeout=840)
    for p in procs:
        p.join(timeout=60)

    import wukong_amd as wk
    from wukong_amd import queries as Q
    from tests.test_dist_gloo import _modifier_plans
    from tests.test_dist_union import _union_plans
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    plans = dict((n, Q.ALL[n]) for n in QUERIES)
    plans.update(_modifier_plans(Q, wk))
    plans.update({f"u_{n}": p for n, p in _union_plans(Q, wk).items()})
    plans.update({f"v_{n}": p for n, p in Q.versatile_plans(full).items()})
    from tests.test_dist_optional import _opt_plans
    plans.update({f"o_{n}": p for n, p in _opt_plans(Q, wk).items()})
    for name, plan in plans.items():
        want = sort_rows(full.run_query(plan))
        assert got[name].shape == want.shape, (name, got[name].shape, want.shape)
        assert np.array_equal(got[name], want), name
