"""Host store build vs oracle store (independent data structures, same
semantics — DESIGN.md §2/§4), generator determinism, partitioning."""
import numpy as np
import pytest

import wukong_amd as wk
from wukong_amd import queries as Q
from tests.oracle_util import OracleCtx


def test_generator_deterministic():
    a = wk.lubm_gen(2, seed=42)
    b = wk.lubm_gen(2, seed=42)
    assert np.array_equal(a, b)
    c = wk.lubm_gen(2, seed=43)
    assert not np.array_equal(a, c)


def test_generator_partition_covers_full():
    full = wk.lubm_gen(2, seed=7)
    parts = [wk.lubm_gen(2, seed=7, sid=r, nsrv=2) for r in range(2)]
    # pso side: triples with s%2==r must all be in partition r
    for r in range(2):
        want = full[full[:, 0] % 2 == r]
        have = parts[r][parts[r][:, 0] % 2 == r]
        sw = set(map(tuple, want.tolist()))
        sh = set(map(tuple, have.tolist()))
        assert sw == sh
    # pos side likewise
    for r in range(2):
        want = full[full[:, 2] % 2 == r]
        have = parts[r][parts[r][:, 2] % 2 == r]
        assert set(map(tuple, want.tolist())) == set(map(tuple, have.tolist()))


def test_store_indexes_match_oracle(lubm4, store4, oracle4):
    for tp in range(14, 25):
        a = np.sort(store4.get_index(tp, wk.DIR_IN))
        b = np.sort(oracle4.get_index(tp, wk.DIR_IN))
        assert np.array_equal(a, b), tp
    for pid in range(2, 14):
        for d in (wk.DIR_IN, wk.DIR_OUT):
            a = np.sort(store4.get_index(pid, d))
            b = np.sort(oracle4.get_index(pid, d))
            assert np.array_equal(a, b), (pid, d)


def test_store_normal_keys_match_oracle(lubm4, store4, oracle4):
    rng = np.random.default_rng(0)
    ids = np.unique(np.concatenate([lubm4[:, 0], lubm4[:, 2]]))
    ids = ids[ids >= (1 << 17)]
    sample = rng.choice(ids, size=min(3000, len(ids)), replace=False)
    for v in sample:
        for pid in range(1, 14):
            for d in (0, 1):
                a = store4.get_triples(int(v), pid, d)
                b = oracle4.get_triples(int(v), pid, d)
                assert np.array_equal(a, b), (v, pid, d)  # exact incl. order


def test_store_edge_lists_sorted(lubm4, store4):
    """k2c/k2k binary search requires ascending edge lists."""
    rng = np.random.default_rng(1)
    ids = np.unique(lubm4[:, 0])
    for v in rng.choice(ids, size=min(500, len(ids)), replace=False):
        for pid in range(1, 14):
            for d in (0, 1):
                e = store4.get_triples(int(v), pid, d)
                if e.size > 1:
                    assert np.all(np.diff(e.astype(np.int64)) > 0)


def test_store_missing_key_empty(store4):
    assert store4.get_triples(12345 + (1 << 17), 13, 0).size == 0
    assert store4.get_index(9999, 0).size == 0


def test_store_partitioned_union(lubm2):
    """2-partition stores jointly hold exactly the 1-partition store's
    edges (per-key union over owners)."""
    full = wk.Store(lubm2, 0, 1)
    p0 = wk.Store(wk.lubm_gen(2, seed=42, sid=0, nsrv=2), 0, 2)
    p1 = wk.Store(wk.lubm_gen(2, seed=42, sid=1, nsrv=2), 1, 2)
    rng = np.random.default_rng(2)
    ids = np.unique(lubm2[:, 0])
    for v in rng.choice(ids, size=200, replace=False):
        owner = [p0, p1][int(v) % 2]
        other = [p0, p1][1 - int(v) % 2]
        for pid in (1, 4, 8):
            a = full.get_triples(int(v), pid, 1)
            b = owner.get_triples(int(v), pid, 1)
            assert np.array_equal(a, b)
            assert other.get_triples(int(v), pid, 1).size == 0


def test_store_integrity_gsck(store4):
    """The reference's gsck full-store scan (gchecker.hpp:364-392)."""
    assert store4.check() == 0


def test_store_integrity_watdiv():
    st = wk.Store(wk.watdiv_gen(500, seed=1))
    assert st.check() == 0


def test_mem_usage(lubm2):
    import wukong_amd as wk
    st = wk.Store(lubm2)
    slots, edges, side = st.mem_usage()
    assert slots == st.num_slots * 16
    assert edges == st.num_edges * 4
    assert side > 0  # type_of + vp CSR + fn maps + type bitmaps


def test_watdiv_generator_partition_covers_full():
    """WatDiv generator partitions like the LUBM one: partition r holds
    every full-gen triple with s%n==r (pso side) and o%n==r (pos side),
    and nothing else relevant (base_loader.hpp:344-352 contract)."""
    full = wk.watdiv_gen(300, seed=7)
    parts = [wk.watdiv_gen(300, seed=7, sid=r, nsrv=3) for r in range(3)]
    for r in range(3):
        want = full[full[:, 0] % 3 == r]
        have = parts[r][parts[r][:, 0] % 3 == r]
        assert set(map(tuple, want.tolist())) == set(map(tuple, have.tolist()))
        want = full[full[:, 2] % 3 == r]
        have = parts[r][parts[r][:, 2] % 3 == r]
        assert set(map(tuple, want.tolist())) == set(map(tuple, have.tolist()))
