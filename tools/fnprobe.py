#!/usr/bin/env python3
"""Focused profile of the functional-map hot steps (Q1: memberOf,
ugDegreeFrom+typeof) — submit path (k_expand_fn) and graph replay
(k_expand_fn_map), plus the classic pipeline for reference.  Run under
rocprofv3 (kernel trace or PMC) to attribute time/traffic per kernel.

WK_UNIV sizes the store (default 2560).  Prints per-step HIP-event
times from the engine's own counters.
"""
import os
import sys
import time

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, HERE)

os.environ.setdefault("WK_KERNEL_TIMING", "1")

import wukong_amd as wk  # noqa: E402
from wukong_amd import queries as Q  # noqa: E402


def delta(a, b, cat):
    return {"usec": round(b[cat]["usec"] - a[cat]["usec"], 1),
            "bytes": b[cat]["bytes"] - a[cat]["bytes"],
            "launches": b[cat]["launches"] - a[cat]["launches"]}


def main():
    nuniv = int(os.environ.get("WK_UNIV", "2560"))
    reps = int(os.environ.get("WK_REPS", "30"))
    t0 = time.time()
    triples = wk.lubm_gen(nuniv, seed=42)
    store = wk.Store(triples)
    del triples
    gstore = wk.GpuStore(store)
    eng = wk.Engine(gstore)
    print(f"setup {time.time()-t0:.1f}s", file=sys.stderr)

    def q1_steps(label):
        s0 = eng.kernel_stats()
        t0 = time.time()
        for _ in range(reps):
            eng.begin_query(Q.ALL["q1"])
            eng.execute_one_pattern()   # i2u grads
            eng.execute_one_pattern()   # k2u memberOf (fn)
            eng.execute_one_pattern()   # k2u ugDegree (+fused typeof) (fn)
            eng.fetch_count()
        s1 = eng.kernel_stats()
        wall = (time.time() - t0) * 1e6 / reps
        out = {c: delta(s0, s1, c) for c in ("probe", "scan", "expand",
                                             "filter", "copy")}
        print(label, f"wall/rep={wall:.0f}us",
              {c: (round(v['usec'] / reps, 1), v['launches'])
               for c, v in out.items()}, flush=True)

    q1_steps("fn-submit")               # k_fn_gather+compact path
    os.environ["WK_FN_DISPATCH"] = "0"
    q1_steps("classic")                 # probe+scan+expand
    os.environ.pop("WK_FN_DISPATCH")

    # whole-q1 EAGER but ASYNC (submit + one sync at fetch): isolates
    # the per-step host-sync gaps from the kernels themselves
    def q1_async(label):
        s0 = eng.kernel_stats()
        t0 = time.time()
        for _ in range(reps):
            eng.submit(Q.ALL["q1"])
            eng.fetch_count()
        wall = (time.time() - t0) * 1e6 / reps
        s1 = eng.kernel_stats()
        print(label, f"wall/rep={wall:.0f}us",
              {c: round((s1[c]['usec'] - s0[c]['usec']) / reps, 1)
               for c in ("probe", "scan", "expand", "filter", "copy")},
              flush=True)

    q1_async("fn-async")
    os.environ["WK_FN_DISPATCH"] = "0"
    q1_async("classic-async")
    os.environ.pop("WK_FN_DISPATCH")

    # graph replay of whole q1 (k_expand_fn_map 1:1 specialization)
    gid = eng.graph_build(Q.ALL["q1"])
    eng.graph_run(gid)
    s0 = eng.kernel_stats()
    t0 = time.time()
    for _ in range(reps):
        eng.graph_launch(gid)
    eng.sync()
    wall = (time.time() - t0) * 1e6 / reps
    s1 = eng.kernel_stats()
    print("graph-q1", f"wall/rep={wall:.0f}us",
          {c: round((s1[c]['usec'] - s0[c]['usec']) / reps, 1)
           for c in ("probe", "scan", "expand", "filter", "copy")},
          flush=True)

    # q7's fn k2k (teacherOf reversed-functional filter) via full q7
    gid7 = eng.graph_build(Q.ALL["q7"])
    eng.graph_run(gid7)
    s0 = eng.kernel_stats()
    t0 = time.time()
    for _ in range(reps):
        eng.graph_launch(gid7)
    eng.sync()
    wall = (time.time() - t0) * 1e6 / reps
    s1 = eng.kernel_stats()
    print("graph-q7", f"wall/rep={wall:.0f}us",
          {c: round((s1[c]['usec'] - s0[c]['usec']) / reps, 1)
           for c in ("probe", "scan", "expand", "filter", "copy")},
          flush=True)


if __name__ == "__main__":
    main()
