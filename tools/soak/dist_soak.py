import os, sys, random
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__)))))
import numpy as np
import torch.multiprocessing as mp


def gen_plans(seed):
    import wukong_amd as wk
    from wukong_amd import Plan, queries as Q
    from tests.test_fuzz_plans import random_plan, PREDS, TYPES
    store = wk.Store(wk.lubm_gen(2, seed=42))
    rng = random.Random(seed)
    plans = []
    while len(plans) < 20:
        base = random_plan(rng, store)
        r = rng.random()
        if r < 0.5:
            plans.append(base)
            continue
        if not all(pp[1] >= 1 for pp in base.patterns):
            continue
        bound = list(base.required_vars); nv = base.nvars
        s = rng.choice(bound); ovar = -(nv + 1)
        if r < 0.75:
            plans.append(Plan(base.patterns, nv + 1, bound + [ovar],
                              unions=[[(s, rng.choice(PREDS), rng.choice([0, 1]), ovar)],
                                      [(s, rng.choice(PREDS), rng.choice([0, 1]), ovar)]]))
        else:
            plans.append(Plan(base.patterns, nv + 1, bound + [ovar],
                              optional=[(s, rng.choice(PREDS), rng.choice([0, 1]), ovar)]))
    return plans


def worker(rank, world, port, seed, results):
    import torch.distributed as dist
    import wukong_amd as wk
    from wukong_amd.dist import DistQuery
    from tests.oracle_util import OracleCtx, OracleExecutor, sort_rows
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        ctxs = [OracleCtx(wk.lubm_gen(2, seed=42, sid=r, nsrv=world),
                          sid=r, nsrv=world) for r in range(world)]
        out = {}
        for j, plan in enumerate(gen_plans(seed)):
            for mode, thr in (("x", 0), ("r", 10**9)):
                ex = OracleExecutor(ctxs[rank], plan, peers=ctxs)
                dq = DistQuery(ex, plan, rank, world, threshold=thr)
                dq.run()
                out[f"{mode}:{j}"] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    seed = int(sys.argv[1])
    port = int(sys.argv[2])
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, 2, port, seed, results))
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
    for j, plan in enumerate(gen_plans(seed)):
        want = sort_rows(full.run_query(plan))
        for mode in ("x", "r"):
            g = got[f"{mode}:{j}"]
            if g.shape != want.shape or not np.array_equal(g, want):
                print("SOAK MISMATCH", seed, mode, j, plan.patterns)
                bad += 1
    print(f"dist soak seed={seed}: bad={bad}")
