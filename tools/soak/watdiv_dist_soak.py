import os, sys, random
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__)))))
import numpy as np
import torch.multiprocessing as mp


def gen_plans(seed):
    import wukong_amd as wk
    from wukong_amd import Plan, queries as Q
    from wukong_amd import watdiv as W
    EDGES = {
        W.T_PRODUCT: [(W.HASGENRE, 1, W.T_GENRE), (W.OFFER_PRODUCT, 0, W.T_OFFER),
                      (W.REVIEW_PRODUCT, 0, W.T_REVIEW), (W.PURCHASED, 0, W.T_USER)],
        W.T_OFFER: [(W.OFFER_PRODUCT, 1, W.T_PRODUCT), (W.RETAILER, 1, W.T_RETAILER)],
        W.T_REVIEW: [(W.REVIEW_PRODUCT, 1, W.T_PRODUCT), (W.REVIEWER, 1, W.T_USER)],
        W.T_USER: [(W.PURCHASED, 1, W.T_PRODUCT), (W.FRIEND, 1, W.T_USER),
                   (W.FRIEND, 0, W.T_USER), (W.REVIEWER, 0, W.T_REVIEW)],
        W.T_GENRE: [(W.HASGENRE, 0, W.T_PRODUCT)],
        W.T_RETAILER: [(W.RETAILER, 0, W.T_OFFER)],
    }
    rng = random.Random(seed)
    plans = []
    for _ in range(15):
        nv = rng.randint(2, 4)
        vars_ = [-(i + 1) for i in range(nv)]
        t = rng.choice(list(EDGES))
        pats = [(t, 1, 0, vars_[0])]
        bound = {vars_[0]: t}
        free = vars_[1:]
        for _ in range(rng.randint(1, 3)):
            s = rng.choice(list(bound))
            pred, d, rt = rng.choice(EDGES[bound[s]])
            if rng.random() < 0.75 and free:
                o = free.pop(0)
                pats.append((s, pred, d, o))
                bound[o] = rt
            else:
                wrong = rng.random() < 0.15
                ft = rng.choice(list(EDGES)) if wrong else bound[s]
                pats.append((s, 1, 1, ft))
        plans.append(Plan(pats, nvars=nv, required_vars=list(bound)))
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
        ctxs = [OracleCtx(wk.watdiv_gen(1500, seed=7, sid=r, nsrv=world),
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
    seed, port, world = int(sys.argv[1]), int(sys.argv[2]), int(sys.argv[3])
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, world, port, seed, results))
             for r in range(world)]
    for p in procs:
        p.start()
    got = results.get(timeout=600)
    for p in procs:
        p.join(timeout=60)
    import wukong_amd as wk
    from tests.oracle_util import OracleCtx, sort_rows
    full = OracleCtx(wk.watdiv_gen(1500, seed=7))
    bad = 0
    for j, plan in enumerate(gen_plans(seed)):
        want = sort_rows(full.run_query(plan))
        for mode in ("x", "r"):
            g = got[f"{mode}:{j}"]
            if g.shape != want.shape or not np.array_equal(g, want):
                print("WD SOAK MISMATCH", seed, mode, j, plan.patterns)
                bad += 1
    print(f"watdiv dist soak seed={seed} world={world}: bad={bad}")
