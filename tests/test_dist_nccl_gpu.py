"""RCCL exchange leg on real hardware: 2 ranks over the nccl backend,
both on cuda:0 (one leased GPU).  This executes the DEVICE exchange path
end-to-end — generate_sub_query split kernels, all_to_all_single of row
chunks over RCCL, load_rbuf_device — the path the 8-GPU scaling bench
uses (reference analog: the GPUDirect-RDMA row-chunk shuffle,
rdma_adaptor.hpp:339-364 + gpu_hash.cu:600-760).  Parity vs the
1-partition oracle.

Two ranks on one device is outside NCCL's support envelope; RCCL
accepts it for this shape.  Every blocking step is bounded so a
regression cannot wedge the box: the test self-kills its workers.
"""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

QUERIES = ["q1", "q2", "q5", "q7"]


def _worker(rank, world, port, results):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch
    import torch.distributed as dist
    import wukong_amd as wk
    from wukong_amd import queries as Q
    from wukong_amd.dist import DistQuery, GpuExecutor
    from tests.oracle_util import sort_rows
    from datetime import timedelta

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("NCCL_DEBUG", "WARN")
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=rank, world_size=world,
                            timeout=timedelta(seconds=120))
    try:
        triples = wk.lubm_gen(2, seed=42, sid=rank, nsrv=world)
        store = wk.Store(triples, sid=rank, nsrv=world)
        eng = wk.Engine(store, device=0)
        out = {}
        for name in QUERIES:
            plan = Q.ALL[name]
            ex = GpuExecutor(eng, plan)
            dq = DistQuery(ex, plan, rank, world, device="cuda:0")
            dq.run()
            merged = dq.gather_result()
            out[name] = sort_rows(merged)
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


def _reap(procs):
    for p in procs:
        p.join(timeout=30)
    for p in procs:
        if p.is_alive():
            p.terminate()
            p.join(timeout=10)
        if p.is_alive():
            p.kill()


@pytest.mark.timeout(600)
def test_rccl_exchange_two_ranks_one_gpu():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29917, results))
             for r in range(2)]
    for p in procs:
        p.start()
    try:
        got = results.get(timeout=420)
    finally:
        _reap(procs)

    import wukong_amd as wk
    from wukong_amd import queries as Q
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    for name in QUERIES:
        want = sort_rows(full.run_query(Q.ALL[name]))
        assert got[name].shape == want.shape, (name, got[name].shape, want.shape)
        assert np.array_equal(got[name], want), name
