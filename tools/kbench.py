"""Per-step kernel roofline measurement via the step API.
Usage: python tools/kbench.py [--univ=2560] q1 q7 q2
Prints, per pattern step: rows in/out and per-category usec/bytes deltas
(device-counted algorithmic bytes, HIP-event usec) -> achieved GB/s.

NB: a fused step consumes TWO plan patterns, so subsequent printed
labels shift by one relative to what actually executed (the engine's
step counter advances by 2; this tool prints plan.patterns[i]).
"""
import os
import sys

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, HERE)
os.environ["WK_KERNEL_TIMING"] = "1"

import wukong_amd as wk  # noqa: E402
from wukong_amd import queries as Q  # noqa: E402


def main():
    univ = 2560
    names = []
    for a in sys.argv[1:]:
        if a.startswith("--univ="):
            univ = int(a.split("=")[1])
        else:
            names.append(a)
    names = names or ["q1", "q7", "q2"]
    triples = wk.lubm_gen(univ, seed=42)
    store = wk.Store(triples)
    del triples
    eng = wk.Engine(store, device=0)

    for name in names:
        plan = Q.ALL[name]
        eng.run_query_count(plan)  # warm
        print(f"=== {name}")
        eng.begin_query(plan)
        prev = eng.kernel_stats()
        rows = 0
        for i, pat in enumerate(plan.patterns):
            n = eng.execute_one_pattern()
            cur = eng.kernel_stats()
            parts = []
            for k in cur:
                du = cur[k]["usec"] - prev[k]["usec"]
                db = cur[k]["bytes"] - prev[k]["bytes"]
                if du > 1:
                    gbs = db / du / 1e3 if du else 0
                    parts.append(f"{k}:{du:.0f}us/{db/1e6:.1f}MB/{gbs:.0f}GBs")
            prev = cur
            print(f"  step{i} {pat} rows {rows}->{n}  " + " ".join(parts),
                  flush=True)
            rows = n


if __name__ == "__main__":
    main()
