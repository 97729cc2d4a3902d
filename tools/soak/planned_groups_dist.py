import os, sys
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__)))))
import numpy as np
import torch.multiprocessing as mp

TEXTS = {
    "u_rev": ("PREFIX ub: <x>\nSELECT ?x ?y WHERE {\n"
              "  ?x rdf:type ub:Department .\n"
              "  { ?y ub:memberOf ?x . } UNION { ?y ub:worksFor ?x . }\n}"),
    "o_rev": ("PREFIX ub: <x>\nSELECT ?x ?y WHERE {\n"
              "  ?x rdf:type ub:UndergraduateStudent .\n"
              "  OPTIONAL { ?y ub:advisor ?x . }\n}"),
    "uo": ("PREFIX ub: <x>\nSELECT ?x ?y ?z WHERE {\n"
           "  ?x rdf:type ub:GraduateStudent .\n"
           "  { ?x ub:memberOf ?y . } UNION { ?y ub:undergraduateDegreeFrom"
           " ?x . }\n  OPTIONAL { ?z ub:advisor ?x . }\n}"),
}

def plans():
    import wukong_amd as wk
    from wukong_amd import planner, sparql
    store = wk.Store(wk.lubm_gen(2, seed=42))
    vocab = sparql.lubm_vocab()
    return {n: planner.plan_text(store, t, vocab) for n, t in TEXTS.items()}

def worker(rank, world, port, results):
    import torch.distributed as dist
    import wukong_amd as wk
    from wukong_amd.dist import DistQuery
    from tests.oracle_util import OracleCtx, OracleExecutor, sort_rows
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        ctx = OracleCtx(wk.lubm_gen(2, seed=42, sid=rank, nsrv=world),
                        sid=rank, nsrv=world)
        out = {}
        for n, plan in plans().items():
            ex = OracleExecutor(ctx, plan)
            dq = DistQuery(ex, plan, rank, world)
            dq.run()
            out[n] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()

if __name__ == "__main__":
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, 2, 29951, results))
             for r in range(2)]
    for p in procs:
        p.start()
    got = results.get(timeout=600)
    for p in procs:
        p.join(timeout=60)
    import wukong_amd as wk
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    bad = 0
    for n, plan in plans().items():
        want = sort_rows(full.run_query(plan))
        if got[n].shape != want.shape or not np.array_equal(got[n], want):
            print("MISMATCH", n, got[n].shape, want.shape); bad += 1
        elif not len(want):
            print("EMPTY", n); bad += 1
    print(f"planned-groups dist soak: bad={bad}")
