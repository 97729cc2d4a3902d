"""Diagnostic: run chosen queries in a loop on one GPU.
Usage: python tools/qloop.py q3 q3 q3 q2 q3 [--univ 2560]
"""
import os
import sys
import time

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, HERE)

import wukong_amd as wk  # noqa: E402
from wukong_amd import queries as Q  # noqa: E402


def main():
    args = [a for a in sys.argv[1:] if not a.startswith("--")]
    univ = 2560
    for a in sys.argv[1:]:
        if a.startswith("--univ="):
            univ = int(a.split("=")[1])
    t0 = time.time()
    triples = wk.lubm_gen(univ, seed=42)
    store = wk.Store(triples)
    del triples
    eng = wk.Engine(store, device=0)
    print(f"setup {time.time()-t0:.1f}s", file=sys.stderr)
    # warm each distinct query once
    for name in dict.fromkeys(args):
        eng.run_query(Q.ALL[name])
    for name in args:
        t = time.time()
        tbl = eng.run_query(Q.ALL[name])
        print(f"{name}: {1e3*(time.time()-t):.2f} ms rows={tbl.shape[0]}",
              file=sys.stderr, flush=True)


if __name__ == "__main__":
    main()
