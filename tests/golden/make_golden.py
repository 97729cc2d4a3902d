"""Regenerate tests/golden/lubm4_golden.json: row counts + order-
independent sha of the sorted result table for Q1-Q7 at LUBM-4 (seed 42),
computed by the ORACLE (pinned itself by brute force + hash goldens).
Run: python tests/golden/make_golden.py"""
import json
import os
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(os.path.dirname(HERE))
sys.path.insert(0, ROOT)

import wukong_amd as wk  # noqa: E402
from wukong_amd import queries as Q  # noqa: E402
from tests.oracle_util import OracleCtx  # noqa: E402
from tests.test_queries_cpu import fnv1a_fast  # noqa: E402


def main():
    triples = wk.lubm_gen(4, seed=42)
    ora = OracleCtx(triples)
    out = {"dataset": "lubm4", "seed": 42, "ntriples": int(triples.shape[0]),
           "queries": {}}
    plans = dict(Q.ALL)
    plans.update(Q.versatile_plans(ora))  # Q8-Q12 (VERSATILE)
    for name, plan in plans.items():
        t = ora.run_query(plan)
        out["queries"][name] = {"rows": int(t.shape[0]), "cols": int(t.shape[1]),
                                "sha": fnv1a_fast(t)}
        print(name, out["queries"][name])
    with open(os.path.join(HERE, "lubm4_golden.json"), "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
