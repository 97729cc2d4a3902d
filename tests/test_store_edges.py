"""Edge cases of the host store build (store.cpp): maximum field
values, duplicate input triples, hub keys and collision chains.

The reference's loader sorts + dedups raw triples before insert
(core/loader/base_loader.hpp:302-373) and its hash table must absorb
arbitrary collision chains via indirect buckets (gstore.hpp:472-540).
These tests pin the same behavior at the domain's boundary values:
vid at the u32 input limit and pid at the 17-bit ikey field limit
(core/store/vertex.hpp:41,88-96).  Conventions (gstore.hpp:61):
dir 0 = IN, 1 = OUT; real vertex ids start at 1<<17 (ids below are
the predicate/type id space); get_index(p, IN) lists p's subjects,
get_index(p, OUT) its objects.
"""
import numpy as np
import pytest

import wukong_amd as wk

IN, OUT = 0, 1
VBASE = 1 << 17              # first valid vertex id (lubm_gen.cpp:86)
MAX_PID = (1 << 17) - 1      # ikey_t pid:17
MAX_VID = (1 << 32) - 1      # u32 input triples


def _edges(st, vid, pid, d):
    return np.sort(st.get_triples(vid, pid, d))


def test_max_id_roundtrip():
    # subjects/objects at the top of the u32 range, predicate at the
    # top of the 17-bit field: pack/hash/segment lookup must all hold.
    # (The id span is sparse, so the optional dense side indexes are
    # skipped and every read goes through the cluster-hash probe.)
    s, p, o = MAX_VID, MAX_PID, MAX_VID - 1
    triples = np.array([
        [s, p, o],
        [s, p, o - 1],
        [o, p, s],          # reverse edge under the same predicate
        [s, 2, 1 << 20],    # small predicate on the same hub subject
    ], dtype=np.uint32)
    st = wk.Store(triples)
    assert st.check() == 0
    assert list(_edges(st, s, p, OUT)) == [o - 1, o]
    assert list(_edges(st, s, p, IN)) == [o]        # o -> s reverse triple
    assert list(_edges(st, o, p, IN)) == [s]
    assert list(_edges(st, o, p, OUT)) == [s]
    assert list(_edges(st, s, 2, OUT)) == [1 << 20]
    # predicate index: IN side lists subjects, OUT side objects
    assert set(st.get_index(p, IN)) == {s, o}
    assert set(st.get_index(p, OUT)) == {o - 1, o, s}
    # absent predicate and absent key both come back empty
    assert st.get_triples(s, 3, OUT).size == 0
    assert st.get_triples(VBASE + 7, p, OUT).size == 0


def test_pid_beyond_field_rejected():
    # pid >= 2^17 cannot be packed into ikey_t -> build refuses
    with pytest.raises(RuntimeError):
        wk.Store(np.array([[VBASE + 1, 1 << 17, VBASE + 2]], dtype=np.uint32))


def test_duplicate_triples_dedupe():
    # each triple repeated 3x in shuffled order == unique input
    rng = np.random.default_rng(7)
    uniq = rng.integers(VBASE, VBASE + 10_000, size=(500, 3), dtype=np.uint32)
    uniq[:, 1] = rng.integers(2, 40, size=500)  # valid pid range
    uniq = np.unique(uniq, axis=0)
    dup = np.repeat(uniq, 3, axis=0)
    rng.shuffle(dup)
    a, b = wk.Store(uniq), wk.Store(dup)
    assert a.check() == 0 and b.check() == 0
    assert a.num_edges == b.num_edges
    assert a.checksum() == b.checksum()
    for v, p in {(int(r[0]), int(r[1])) for r in uniq[:50]}:
        assert np.array_equal(_edges(a, v, p, OUT), _edges(b, v, p, OUT))


def test_hub_vertex_and_collision_chains():
    # one hub subject: 5000 objects on one predicate (long edge list)
    # plus 3000 distinct predicates with one edge each (3000 distinct
    # keys for the same vid -> collision chains in many 1-key segments).
    hub = 99_999_999
    p_big = 2
    objs = np.arange(1 << 18, (1 << 18) + 5000, dtype=np.uint32)
    t1 = np.stack([np.full(5000, hub, np.uint32),
                   np.full(5000, p_big, np.uint32), objs], axis=1)
    pids = np.arange(3, 3003, dtype=np.uint32)
    t2 = np.stack([np.full(3000, hub, np.uint32), pids,
                   np.full(3000, 1 << 19, np.uint32)], axis=1)
    st = wk.Store(np.vstack([t1, t2]))
    assert st.check() == 0
    assert np.array_equal(_edges(st, hub, p_big, OUT), objs)
    # every 1-edge predicate resolves through its own segment
    for p in (3, 1000, 3002):
        assert list(st.get_triples(hub, p, OUT)) == [1 << 19]
        assert list(st.get_index(p, IN)) == [hub]
        assert list(st.get_triples(1 << 19, p, IN)) == [hub]
    k, e = st.seg_stats(p_big, OUT)
    assert (k, e) == (1, 5000)
    k, e = st.seg_stats(p_big, IN)
    assert (k, e) == (5000, 5000)


def test_singleton_and_absent():
    s, o = VBASE + 10, VBASE + 20
    st = wk.Store(np.array([[s, 2, o]], dtype=np.uint32))
    assert st.check() == 0
    assert list(st.get_triples(s, 2, OUT)) == [o]
    assert list(st.get_triples(o, 2, IN)) == [s]
    assert list(st.get_index(2, IN)) == [s]
    assert list(st.get_index(2, OUT)) == [o]
    assert st.get_index(5, IN).size == 0
    assert st.get_triples(s, 2, IN).size == 0  # wrong direction is empty


def test_random_triples_vs_brute():
    """Randomized roundtrip: every (s,p,OUT)/(o,p,IN) list and both
    index sides must equal a brute-force dict over the same triples."""
    rng = np.random.default_rng(123)
    n = 4000
    t = np.stack([
        rng.integers(VBASE, VBASE + 300, size=n, dtype=np.uint32),
        rng.integers(2, 25, size=n, dtype=np.uint32),
        rng.integers(VBASE, VBASE + 300, size=n, dtype=np.uint32),
    ], axis=1)
    t = np.unique(t, axis=0)
    st = wk.Store(t)
    assert st.check() == 0
    out_d, in_d, subj, obj = {}, {}, {}, {}
    for s, p, o in t:
        out_d.setdefault((int(s), int(p)), set()).add(int(o))
        in_d.setdefault((int(o), int(p)), set()).add(int(s))
        subj.setdefault(int(p), set()).add(int(s))
        obj.setdefault(int(p), set()).add(int(o))
    for (v, p), want in out_d.items():
        assert list(_edges(st, v, p, OUT)) == sorted(want)
    for (v, p), want in in_d.items():
        assert list(_edges(st, v, p, IN)) == sorted(want)
    for p in subj:
        assert set(st.get_index(p, IN)) == subj[p]
        assert set(st.get_index(p, OUT)) == obj[p]
        assert st.seg_stats(p, OUT) == (len(subj[p]),
                                        sum(len(out_d[k]) for k in out_d if k[1] == p))


def test_partition_union_random_nsrv3():
    """Random triples partitioned 3 ways: every partition holds exactly
    its owned rows (pso by s%3, pos by o%3 — base_loader.hpp:344-352)
    and the partitions' OUT edge lists union to the full store's."""
    rng = np.random.default_rng(99)
    t = np.stack([
        rng.integers(VBASE, VBASE + 500, size=3000, dtype=np.uint32),
        rng.integers(2, 20, size=3000, dtype=np.uint32),
        rng.integers(VBASE, VBASE + 500, size=3000, dtype=np.uint32),
    ], axis=1)
    t = np.unique(t, axis=0)
    full = wk.Store(t)
    parts = [wk.Store(t, sid=r, nsrv=3) for r in range(3)]
    assert all(p.check() == 0 for p in parts)
    for v, p in {(int(r[0]), int(r[1])) for r in t[:200]}:
        want = _edges(full, v, p, OUT)
        got = _edges(parts[v % 3], v, p, OUT)
        assert np.array_equal(got, want), (v, p)
        for r in range(3):  # non-owners hold nothing for this key
            if r != v % 3:
                assert parts[r].get_triples(v, p, OUT).size == 0
    # subject indexes union (with cross-partition dedupe) to the full set
    for p in range(2, 20):
        want = set(full.get_index(p, IN))
        got = set()
        for r in range(3):
            got |= set(parts[r].get_index(p, IN))
        assert got == want, p


def test_plan_rejects_out_of_range_vars():
    """A var id below -nvars would index past the engine's v2c array on
    the C side — the binding must reject it before it crosses the ABI."""
    with pytest.raises(ValueError):
        wk.Plan([(2, 1, IN, -4)], nvars=2, required_vars=[-1])
    with pytest.raises(ValueError):
        wk.Plan([(2, 1, IN, -1)], nvars=1, required_vars=[-2])
    with pytest.raises(ValueError):
        wk.Plan([(2, 1, IN, -1)], nvars=2, required_vars=[-1],
                unions=[[(-1, 2, OUT, -5)]])
