import os, sys, random
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__)))))
import numpy as np
import torch.multiprocessing as mp


def gen_plans(seed):
    """Optional groups with up to 3 patterns: k2u then filters/k2k,
    including const filters and typeof checks on group-born columns."""
    import wukong_amd as wk
    from wukong_amd import Plan, queries as Q
    from tests.test_fuzz_plans import random_plan, PREDS, TYPES
    store = wk.Store(wk.lubm_gen(2, seed=42))
    rng = random.Random(seed)
    plans = []
    while len(plans) < 14:
        base = random_plan(rng, store)
        if not all(pp[1] >= 1 for pp in base.patterns):
            continue
        bound = list(base.required_vars); nv = base.nvars
        s = rng.choice(bound)
        o1 = -(nv + 1)
        group = [(s, rng.choice(PREDS), rng.choice([0, 1]), o1)]
        nvx = nv + 1
        r = rng.random()
        if r < 0.3:
            group.append((o1, Q.TYPE_ID, 1, rng.choice(TYPES)))
            group.append((o1, rng.choice(PREDS), rng.choice([0, 1]),
                          rng.choice(bound)))
        elif r < 0.6:
            o2 = -(nv + 2); nvx = nv + 2
            group.append((o1, rng.choice(PREDS), rng.choice([0, 1]), o2))
            group.append((o2, Q.TYPE_ID, 1, rng.choice(TYPES)))
        else:
            group.append((Q.DEPT0_UNIV0, rng.choice([Q.WORKSFOR, Q.MEMBEROF]),
                          0, o1))
        plans.append(Plan(base.patterns, nvx, bound + [o1], optional=group))
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
        ctx = OracleCtx(wk.lubm_gen(2, seed=42, sid=rank, nsrv=world),
                        sid=rank, nsrv=world)
        out = {}
        for j, plan in enumerate(gen_plans(seed)):
            ex = OracleExecutor(ctx, plan)
            dq = DistQuery(ex, plan, rank, world)
            dq.run()
            out[j] = sort_rows(dq.gather_result())
        if rank == 0:
            results.put(out)
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    seed, port, world = int(sys.argv[1]), int(sys.argv[2]), int(sys.argv[3])
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, world, port, seed, results))
             for r in range(world)]
    for p in procs:
        p.start()
    got = results.get(timeout=900)
    for p in procs:
        p.join(timeout=60)
    import wukong_amd as wk
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.lubm_gen(2, seed=42))
    bad = 0
    for j, plan in enumerate(gen_plans(seed)):
        want = sort_rows(full.run_query(plan))
        if got[j].shape != want.shape or not np.array_equal(got[j], want):
            print("OPT3 MISMATCH", seed, j, plan.patterns, plan.optional,
                  got[j].shape, want.shape)
            bad += 1
    print(f"opt3 dist soak seed={seed} world={world}: bad={bad}")
