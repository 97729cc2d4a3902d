import os, sys, random
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__)))))
import numpy as np
import torch.multiprocessing as mp


def gen_plans(seed):
    import wukong_amd as wk
    from wukong_amd import planner
    from tests.test_fuzz_plans import random_plan
    store = wk.Store(wk.lubm_gen(2, seed=42))
    rng = random.Random(seed)
    plans = []
    while len(plans) < 20:
        base = random_plan(rng, store)
        shuffled = list(base.patterns)
        rng.shuffle(shuffled)
        try:
            planned = planner.plan_patterns(store, shuffled, base.nvars,
                                            base.required_vars)
        except planner.PlannerError:
            continue
        plans.append((base, planned))
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
        for j, (base, planned) in enumerate(gen_plans(seed)):
            for mode, thr in (("x", 0), ("m", 300)):
                ex = OracleExecutor(ctxs[rank], planned, peers=ctxs)
                dq = DistQuery(ex, planned, rank, world, threshold=thr)
                dq.run()
                out[f"{mode}:{j}"] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    seed, port = int(sys.argv[1]), int(sys.argv[2])
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, 2, port, seed, results))
             for r in range(2)]
    for p in procs:
        p.start()
    got = results.get(timeout=900)
    for p in procs:
        p.join(timeout=60)
    import wukong_amd as wk
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    bad = 0
    for j, (base, planned) in enumerate(gen_plans(seed)):
        want = sort_rows(full.run_query(base))  # textual-order truth
        for mode in ("x", "m"):
            g = got[f"{mode}:{j}"]
            if g.shape != want.shape or not np.array_equal(g, want):
                print("PLANNED-DIST MISMATCH", seed, mode, j,
                      base.patterns, planned.patterns)
                bad += 1
    print(f"planned-dist soak seed={seed}: bad={bad}")
