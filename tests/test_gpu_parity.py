"""GPU parity — the first gate: the HIP engine's bindings
must equal the oracle's, set-equal, on the same inputs.  All tests here
need a real MI355X (`-m gpu`; /root/reference is NOT read)."""
import json
import os

import numpy as np
import pytest

import wukong_amd as wk
from wukong_amd import Plan, queries as Q
from tests.oracle_util import OracleCtx, sort_rows

pytestmark = pytest.mark.gpu

GOLD = os.path.join(os.path.dirname(__file__), "golden", "lubm4_golden.json")


@pytest.fixture(scope="module")
def eng4(store4):
    return wk.Engine(store4, device=0)


@pytest.mark.parametrize("name", list(Q.ALL))
def test_query_parity_lubm4(name, eng4, oracle4):
    plan = Q.ALL[name]
    got = eng4.run_query(plan)
    want = oracle4.run_query(plan)
    assert got.shape == want.shape, (got.shape, want.shape)
    assert np.array_equal(sort_rows(got), sort_rows(want))


def test_side_index_fallback_parity(lubm4, oracle4):
    """Pure cluster-hash path: a store built WITHOUT the side indexes
    (fn maps, CSR, type bitmaps — WK_FN=0 WK_CSR=0 WK_TBM=0) must give
    identical bindings.  Covers the probe/expand/filter base kernels
    that the side-indexed default store routes around."""
    old = {}
    for k in ("WK_FN", "WK_CSR", "WK_TBM"):
        old[k] = os.environ.get(k)
        os.environ[k] = "0"
    try:
        store = wk.Store(lubm4)
        eng = wk.Engine(store, device=0)
        for name, plan in Q.ALL.items():
            got = sort_rows(eng.run_query(plan))
            want = sort_rows(oracle4.run_query(plan))
            assert got.shape == want.shape, (name, got.shape, want.shape)
            assert np.array_equal(got, want), name
    finally:
        for k, v in old.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v


def test_golden_fixture_gpu(eng4, store4):
    if not os.path.exists(GOLD):
        pytest.skip("golden fixture not generated")
    from tests.test_queries_cpu import fnv1a_fast
    with open(GOLD) as f:
        gold = json.load(f)
    plans = dict(Q.ALL)
    plans.update(Q.versatile_plans(store4))
    for name, rec in gold["queries"].items():
        t = eng4.run_query(plans[name])
        assert t.shape[0] == rec["rows"], name
        assert fnv1a_fast(t) == rec["sha"], name


def test_distinct_limit_gpu(eng4, oracle4):
    p = Plan(Q.Q2.patterns, Q.Q2.nvars, [Q.X], distinct=True)
    got = eng4.run_query(p)
    want = oracle4.run_query(p)
    assert np.array_equal(sort_rows(got), sort_rows(want))
    p2 = Plan(Q.Q2.patterns, Q.Q2.nvars, Q.Q2.required_vars, limit=13)
    assert eng4.run_query(p2).shape[0] == 13


def test_step_api_and_split(store4, oracle4):
    """Step-level API + generate_sub_query chunks (fork-join split)."""
    eng = wk.Engine(store4, device=0)
    plan = Q.Q1
    eng.begin_query(plan)
    n0 = eng.execute_one_pattern()   # i2u
    n1 = eng.execute_one_pattern()   # k2u memberOf
    assert n1 > 0
    # split the current table by next pattern's start var (?X, col 0)
    ncols = eng.col_num
    buf = wk.dev_alloc(n1 * ncols * 4)
    rows = eng.generate_sub_query(2, buf, n1)
    assert sum(rows) == n1
    tbl = eng.fetch_raw()
    packed = wk.dev_download_u32(buf, n1 * ncols).reshape(n1, ncols)
    wk.dev_free(buf)
    # chunk d rows all hash to d; union of chunks == table (multiset)
    xcol = 0
    assert np.all(packed[: rows[0], xcol] % 2 == 0)
    assert np.all(packed[rows[0]:, xcol] % 2 == 1)
    assert np.array_equal(sort_rows(packed), sort_rows(tbl))


def test_parity_lubm40_counts():
    """Bigger store: Q1-Q7 row counts + checksums vs oracle at LUBM-40."""
    triples = wk.lubm_gen(40, seed=42)
    store = wk.Store(triples)
    eng = wk.Engine(store, device=0)
    ora = OracleCtx(triples)
    for name, plan in Q.ALL.items():
        got = eng.run_query(plan)
        want = ora.run_query(plan)
        assert got.shape == want.shape, (name, got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), name


def test_empty_and_edge_cases(eng4):
    # const with no edges -> empty, then ops on empty stay empty
    p = Plan([(Q.UNIV0 + 1, Q.TAKESCOURSE, wk.DIR_OUT, -1),
              (-1, Q.TYPE_ID, wk.DIR_OUT, Q.COURSE)],
             nvars=1, required_vars=[-1])
    assert eng4.run_query(p).shape[0] == 0
    # k2u producing a new column on empty table keeps col bookkeeping
    p2 = Plan([(Q.UNIV0 + 1, Q.TAKESCOURSE, wk.DIR_OUT, -1),
               (-1, Q.NAME, wk.DIR_OUT, -2)],
              nvars=2, required_vars=[-1, -2])
    t = eng4.run_query(p2)
    assert t.shape == (0, 2)


def test_overflow_rerun(lubm4, oracle4):
    """Force tiny capacities so every heavy step overflows: the device
    flag + grow + re-run path must still produce exact results (replaces
    the reference's hard rbuf assert, gpu_engine_cuda.hpp:185)."""
    import subprocess
    import sys
    code = """
import os, sys
os.environ['WK_MIN_CAP'] = '1000'
sys.path.insert(0, %r)
import numpy as np
import wukong_amd as wk
from wukong_amd import queries as Q
from tests.oracle_util import OracleCtx, sort_rows
triples = wk.lubm_gen(4, seed=42)
store = wk.Store(triples)
eng = wk.Engine(store, device=0)
ora = OracleCtx(triples)
for name in ('q1', 'q2', 'q7'):
    got = eng.run_query(Q.ALL[name])
    want = ora.run_query(Q.ALL[name])
    assert got.shape == want.shape, (name, got.shape, want.shape)
    assert np.array_equal(sort_rows(got), sort_rows(want)), name
# step API path (mid-plan overflow re-run)
eng.begin_query(Q.ALL['q1'])
for _ in Q.ALL['q1'].patterns:
    eng.execute_one_pattern()
got = sort_rows(eng.fetch_result())
assert np.array_equal(got, sort_rows(ora.run_query(Q.ALL['q1'])))
print('OVERFLOW_OK')
""" % (os.path.dirname(os.path.dirname(os.path.abspath(__file__))),)
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=300)
    assert "OVERFLOW_OK" in r.stdout, r.stdout + r.stderr


def test_emulator_templates_parity(lubm4, store4, oracle4):
    """Light emulator templates (A1-A6) through the async submit path —
    including the single-kernel light-query fast path — vs the oracle."""
    import random
    eng = wk.Engine(store4, device=0)
    rng = random.Random(3)
    for tname in Q.EMU_WEIGHTS:
        pool = store4.get_index(Q.EMU_POOLS[tname], wk.DIR_IN)
        assert pool.size > 0, tname
        for _ in range(4):
            const = int(rng.choice(pool))
            plan = Q.emu_template(tname, const)
            eng.submit(plan)
            got = eng.fetch_result()
            want = oracle4.run_query(plan)
            assert got.shape == want.shape, (tname, const, got.shape, want.shape)
            assert np.array_equal(sort_rows(got), sort_rows(want)), (tname, const)


def test_light_batch_parity(store4, oracle4):
    """Batched light-query window (wk_engine_submit_light_batch — one
    launch per window, one wavefront workgroup per query) vs per-query
    oracle counts, including empty-pool constants and double-buffered
    back-to-back windows."""
    import random
    eng = wk.Engine(store4, device=0)
    rng = random.Random(11)
    light = {}
    for tname in Q.EMU_WEIGHTS:
        p = Q.emu_template(tname, 1)
        if len(p.patterns) == 2 and p.patterns[1][1] == Q.TYPE_ID:
            light[tname] = (p.patterns[0][1], p.patterns[0][2],
                            p.patterns[1][3])
    assert set(light) == {"a1", "a2", "a3", "a5"}
    for trial in range(3):
        subj, pred, dirs, cval, plans = [], [], [], [], []
        for _ in range(37 + trial):
            tname = rng.choice(list(light))
            pool = store4.get_index(Q.EMU_POOLS[tname], wk.DIR_IN)
            const = int(rng.choice(pool))
            pr, dr, cv = light[tname]
            subj.append(const)
            pred.append(pr)
            dirs.append(dr)
            cval.append(cv)
            plans.append(Q.emu_template(tname, const))
        # one nonexistent constant: must count 0, not crash
        subj.append(3)
        pred.append(Q.TAKESCOURSE)
        dirs.append(wk.DIR_IN)
        cval.append(Q.GRADSTUDENT)
        plans.append(None)
        eng.submit_light_batch(subj, pred, dirs, cval)
        counts = eng.wait_light_batch()
        for i, plan in enumerate(plans):
            want = 0 if plan is None else len(oracle4.run_query(plan))
            assert counts[i] == want, (trial, i, counts[i], want)


def test_plan_batch_parity(store4, oracle4):
    """LDS plan interpreter (wk_engine_submit_plan_batch): whole
    multi-pattern templates batched one workgroup per query — counts vs
    the oracle, including a nonexistent constant."""
    import random
    eng = wk.Engine(store4, device=0)
    rng = random.Random(5)
    for tname in ["a4", "a6", "a1"]:
        pool = store4.get_index(Q.EMU_POOLS[tname], wk.DIR_IN)
        consts = [int(rng.choice(pool)) for _ in range(23)]
        consts.append(1 << 17)  # vid likely without this predicate
        tmpl = Q.emu_template(tname, 1 << 17)
        eng.submit_plan_batch(tmpl, consts)
        counts = eng.wait_light_batch()
        assert not np.any(counts == np.uint64(wk.LP_OVERFLOW)), tname
        for i, c in enumerate(consts):
            want = len(oracle4.run_query(Q.emu_template(tname, c)))
            assert counts[i] == want, (tname, c, counts[i], want)


def test_plan_batch_rejects_uninterpretable(store4):
    """Index-start plans (Q1) are not LDS-interpretable — must raise
    (fallback contract), not mis-execute."""
    eng = wk.Engine(store4, device=0)
    with pytest.raises(ValueError):
        eng.submit_plan_batch(Q.Q1, [1 << 17])


def test_graph_replay_counts(store4, oracle4):
    """hipGraph capture/replay (wk_engine_graph_build/run): one
    hipGraphLaunch per whole query, counts equal the oracle across
    repeated replays (device state resets inside the graph)."""
    eng = wk.Engine(store4, device=0)
    for name, plan in Q.ALL.items():
        gid = eng.graph_build(plan)
        want = len(oracle4.run_query(plan))
        for _ in range(3):
            assert eng.graph_run(gid) == want, name


def test_graph_suite_replay(store4, oracle4):
    """Whole-suite capture (wk_engine_graph_build_suite): all plans in
    ONE graph; replays stay parity-green — the LAST query's count is
    published (graph_run semantics), and the sticky overflow guard
    covers every query inside the capture (eng.sync raises on S_ERR)."""
    eng = wk.Engine(store4, device=0)
    plans = [Q.ALL[n] for n in ("q1", "q2", "q5", "q7")]
    gid = eng.graph_build_suite(plans)
    want_last = len(oracle4.run_query(plans[-1]))
    for _ in range(3):
        assert eng.graph_run(gid) == want_last
    for _ in range(5):  # async replays, one sync (the timed-loop shape)
        eng.graph_launch(gid)
    eng.sync()
