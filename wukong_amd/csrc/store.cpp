/*
 * store.cpp — host-side build of the HBM-resident predicate-segmented
 * cluster-hash store.
 *
 * Restates (not copies) the reference build pipeline:
 *  - partition + sort pso/pos + dedup: core/loader/base_loader.hpp:302-373
 *  - per-segment key/edge insertion:   core/store/static_gstore.hpp:64-161
 *  - index collection + insertion:     core/store/gstore.hpp:858-888,
 *                                      static_gstore.hpp:217-280
 *  - cluster-hash insert/probe:        core/store/gstore.hpp:789-856,341-361
 * Bucket allocation policy (per-segment ceil(keys/5) main buckets + shared
 * ext region) is ours — DESIGN.md §2.
 */
#include "wk_store.h"
#include "../../include/wukong_abi.h"
#include <algorithm>
#include <atomic>
#include <cstring>
#include <cstdio>
#include <chrono>
#include <omp.h>

static double now_s() {
    return std::chrono::duration<double>(
               std::chrono::steady_clock::now().time_since_epoch()).count();
}
static bool wk_verbose() {
    static int v = -1;
    if (v < 0) { const char *e = getenv("WK_VERBOSE"); v = e && atoi(e); }
    return v;
}
#define WK_LOG(...) do { if (wk_verbose()) { fprintf(stderr, __VA_ARGS__); } } while (0)

using namespace wk;

namespace {

struct triple { sid_t s, p, o; };

// parallel counting sort into buckets keyed by (p, id>>17), then per-bucket
// std::sort — preserves full (p, major, minor) order because id>>17 is the
// major prefix of the secondary key.
static void sort_triples(std::vector<triple> &t, bool by_s /* pso vs pos */,
                         uint32_t max_pid, uint32_t max_blk) {
    const uint64_t nb = (uint64_t)(max_pid + 1) * (max_blk + 2);
    const int64_t n = (int64_t)t.size();
    // the bucket pass needs nthreads*nb counters — only worth it (and
    // only affordable) when the (pid, vid-block) grid is small next to
    // the input; tiny or id-sparse inputs take the direct sort.
    if (nb > (uint64_t)n / 4 + 4096) {
        auto full_pso = [](const triple &a, const triple &b) {
            return a.p != b.p ? a.p < b.p : a.s != b.s ? a.s < b.s : a.o < b.o;
        };
        auto full_pos = [](const triple &a, const triple &b) {
            return a.p != b.p ? a.p < b.p : a.o != b.o ? a.o < b.o : a.s < b.s;
        };
        if (by_s) std::sort(t.begin(), t.end(), full_pso);
        else      std::sort(t.begin(), t.end(), full_pos);
        return;
    }
    auto bucket_of = [&](const triple &x) -> uint64_t {
        uint32_t major = by_s ? x.s : x.o;
        return (uint64_t)x.p * (max_blk + 2) + (major >> NBITS_IDX);
    };
    int nthr = omp_get_max_threads();
    std::vector<uint64_t> hist((size_t)nthr * nb, 0);
#pragma omp parallel num_threads(nthr)
    {
        int tid = omp_get_thread_num();
        uint64_t *h = hist.data() + (size_t)tid * nb;
#pragma omp for schedule(static)
        for (int64_t i = 0; i < n; i++) h[bucket_of(t[i])]++;
    }
    // exclusive offsets: bucket-major, thread-minor
    std::vector<uint64_t> bucket_off(nb + 1, 0);
    {
        uint64_t acc = 0;
        for (uint64_t b = 0; b < nb; b++) {
            bucket_off[b] = acc;
            for (int k = 0; k < nthr; k++) {
                uint64_t c = hist[(size_t)k * nb + b];
                hist[(size_t)k * nb + b] = acc;
                acc += c;
            }
        }
        bucket_off[nb] = acc;
    }
    std::vector<triple> out(t.size());
#pragma omp parallel num_threads(nthr)
    {
        int tid = omp_get_thread_num();
        uint64_t *h = hist.data() + (size_t)tid * nb;
#pragma omp for schedule(static)
        for (int64_t i = 0; i < n; i++) out[h[bucket_of(t[i])]++] = t[i];
    }
    t.swap(out);
    // per-bucket comparison sort on the minor keys
    auto cmp_pso = [](const triple &a, const triple &b) {
        return a.s != b.s ? a.s < b.s : a.o < b.o;
    };
    auto cmp_pos = [](const triple &a, const triple &b) {
        return a.o != b.o ? a.o < b.o : a.s < b.s;
    };
#pragma omp parallel for schedule(dynamic, 16)
    for (uint64_t b = 0; b < nb; b++) {
        auto *lo = t.data() + bucket_off[b], *hi = t.data() + bucket_off[b + 1];
        if (hi - lo > 1) {
            if (by_s) std::sort(lo, hi, cmp_pso);
            else      std::sort(lo, hi, cmp_pos);
        }
    }
}

static void dedup_triples(std::vector<triple> &t) {
    auto eq = [](const triple &a, const triple &b) {
        return a.s == b.s && a.p == b.p && a.o == b.o;
    };
    auto end = std::unique(t.begin(), t.end(), eq);
    t.erase(end, t.end());
}

// cluster-hash inserter for one segment (single-threaded per segment;
// segments insert in parallel).  Mirrors GStore::insert_key
// (gstore.hpp:789-856) with ext buckets drawn from a shared atomic region.
struct inserter {
    wk_store &st;
    const seg_t &seg;
    std::atomic<uint64_t> &ext_next;

    uint64_t insert(uint64_t key) {
        uint64_t bucket = seg.bucket_start + hash_u64(key) % seg.num_buckets;
        while (true) {
            vertex_t *b = &st.vertices[bucket * ASSOC];
            for (int i = 0; i < ASSOC - 1; i++) {
                if (b[i].key == key) { fprintf(stderr, "dup key!\n"); abort(); }
                if (b[i].key == KEY_EMPTY) { b[i].key = key; return bucket * ASSOC + i; }
            }
            if (b[ASSOC - 1].key != KEY_EMPTY) {
                bucket = key_vid(b[ASSOC - 1].key);
                continue;
            }
            uint64_t nb = ext_next.fetch_add(1);
            if (nb >= st.nbuckets_main + st.nbuckets_ext) {
                fprintf(stderr, "out of ext buckets\n"); abort();
            }
            b[ASSOC - 1].key = key_pack(nb, 0, 0);
            bucket = nb;
        }
    }
};

}  // namespace

static wk_store_t *store_build_impl(const sid_t *spo, int64_t ntriples,
                                    int32_t sid, int32_t nsrv) {
    if (!spo || ntriples < 0 || nsrv <= 0 || sid < 0 || sid >= nsrv) return nullptr;
    wk_store *st = new wk_store();
    st->sid = sid; st->nsrv = nsrv;
    double t0 = now_s();

    // 1. partition into pso (s%n==sid) / pos (o%n==sid) — base_loader.hpp:344-352
    std::vector<triple> pso, pos;
    pso.reserve(ntriples); pos.reserve(ntriples);
    uint32_t max_pid = 1, max_id = 0;
    for (int64_t i = 0; i < ntriples; i++) {
        triple t{spo[3 * i], spo[3 * i + 1], spo[3 * i + 2]};
        max_pid = std::max(max_pid, t.p);
        if (t.p == TYPE_ID) max_pid = std::max(max_pid, t.o);  // types are index ids
        max_id = std::max({max_id, t.s, t.o});
        if ((int32_t)(t.s % (sid_t)nsrv) == sid) pso.push_back(t);
        if ((int32_t)(t.o % (sid_t)nsrv) == sid) pos.push_back(t);
    }
    if (max_pid >= (1u << NBITS_IDX)) { delete st; return nullptr; }
    st->max_pid = max_pid;
    uint32_t max_blk = max_id >> NBITS_IDX;
    // The dense side indexes (type_of, VERSATILE vp-CSR) are sized by
    // the vid SPAN, not the triple count.  The generators allocate vids
    // contiguously from 1<<17 (lubm_gen.cpp:86), so span >> input only
    // happens on hand-built input with ids parked near 2^32 — there the
    // dense arrays would cost GBs for a handful of triples.  Skip the
    // optional indexes in that case; the hash-probe fallback stays
    // correct (fn/CSR/type-bitmap builds are already byte-budgeted).
    const uint64_t vid_span = max_id >= (1u << NBITS_IDX)
                                  ? (uint64_t)max_id + 1 - (1u << NBITS_IDX)
                                  : 0;
    const bool dense_vids = vid_span <= 16ull * (uint64_t)ntriples + (1u << 22);

    WK_LOG("[store] partition: %.1fs (pso=%zu pos=%zu)\n", now_s() - t0, pso.size(), pos.size());
    t0 = now_s();
    // 2. sort + dedup — base_loader.hpp:367-377
    sort_triples(pso, true, max_pid, max_blk);
    sort_triples(pos, false, max_pid, max_blk);
    WK_LOG("[store] sort: %.1fs\n", now_s() - t0); t0 = now_s();
    dedup_triples(pso);
    dedup_triples(pos);
    WK_LOG("[store] dedup: %.1fs\n", now_s() - t0); t0 = now_s();

    // 3. pid slice boundaries in each sorted array
    const uint32_t NP = max_pid + 1;
    std::vector<int64_t> pso_lo(NP + 1, 0), pos_lo(NP + 1, 0);
    {
        int64_t i = 0;
        for (uint32_t p = 0; p <= max_pid; p++) {
            pso_lo[p] = i;
            while (i < (int64_t)pso.size() && pso[i].p == p) i++;
        }
        pso_lo[NP] = (int64_t)pso.size();
        // pso is bucket-sorted by p ascending, so slices are contiguous —
        // but only if every element was consumed:
        if (i != (int64_t)pso.size()) { fprintf(stderr, "pso slice error\n"); abort(); }
        i = 0;
        for (uint32_t p = 0; p <= max_pid; p++) {
            pos_lo[p] = i;
            while (i < (int64_t)pos.size() && pos[i].p == p) i++;
        }
        pos_lo[NP] = (int64_t)pos.size();
        if (i != (int64_t)pos.size()) { fprintf(stderr, "pos slice error\n"); abort(); }
    }

    WK_LOG("[store] slices: %.1fs\n", now_s() - t0); t0 = now_s();
    // 4. count keys/edges per segment.
    //    OUT seg of pid: (s,p) runs in pso slice.
    //    IN  seg of pid: (o,p) runs in pos slice, skipping is_tpid(o) runs
    //    (type triples are never normal IN keys — static_gstore.hpp:128-131).
    std::vector<uint64_t> out_keys(NP, 0), out_edges(NP, 0), in_keys(NP, 0), in_edges(NP, 0);
#pragma omp parallel for schedule(dynamic)
    for (uint32_t p = 0; p <= max_pid; p++) {
        for (int64_t i = pso_lo[p]; i < pso_lo[p + 1];) {
            int64_t j = i + 1;
            while (j < pso_lo[p + 1] && pso[j].s == pso[i].s) j++;
            out_keys[p]++; out_edges[p] += (uint64_t)(j - i);
            i = j;
        }
        for (int64_t i = pos_lo[p]; i < pos_lo[p + 1];) {
            int64_t j = i + 1;
            while (j < pos_lo[p + 1] && pos[j].o == pos[i].o) j++;
            if (!is_tpid(pos[i].o)) { in_keys[p]++; in_edges[p] += (uint64_t)(j - i); }
            i = j;
        }
    }

    st->seg_keys.assign((size_t)NP * 2, 0);
    st->seg_edges.assign((size_t)NP * 2, 0);
    for (uint32_t p = 0; p <= max_pid; p++) {
        st->seg_keys[p * 2 + DIR_OUT] = out_keys[p];
        st->seg_edges[p * 2 + DIR_OUT] = out_edges[p];
        st->seg_keys[p * 2 + DIR_IN] = in_keys[p];
        st->seg_edges[p * 2 + DIR_IN] = in_edges[p];
    }
    WK_LOG("[store] seg counts: %.1fs\n", now_s() - t0); t0 = now_s();
    // index segment contents (gstore.hpp:858-888):
    //   [0|pid|IN]  = subjects of pid (from OUT keys, pid != TYPE_ID)
    //   [0|pid|OUT] = objects  of pid (from IN keys)
    //   [0|tid|IN]  = type members (from [vid|TYPE_ID|OUT] edges)
    uint64_t idx_in_keys = 0, idx_in_edges = 0, idx_out_keys = 0, idx_out_edges = 0;
    std::vector<uint64_t> type_members(NP, 0);
    for (int64_t i = pso_lo[TYPE_ID]; i < pso_lo[TYPE_ID + 1]; i++)
        type_members[pso[i].o]++;
    for (uint32_t p = 2; p <= max_pid; p++) {
        if (out_keys[p]) { idx_in_keys++; idx_in_edges += out_keys[p]; }
        if (in_keys[p])  { idx_out_keys++; idx_out_edges += in_keys[p]; }
        if (type_members[p]) { idx_in_keys++; idx_in_edges += type_members[p]; }
    }
    // [vid|TYPE_ID|OUT] keys exist but contribute nothing to pidx maps
    // (collect_idx_info, gstore.hpp:873-887)

    // 5. allocate segments: buckets + edge offsets
    st->nseg.assign((size_t)NP * 2, seg_t{});
    uint64_t bucket_cursor = 0, edge_cursor = 0;
    std::vector<uint64_t> seg_edge_start((size_t)NP * 2, 0);
    uint64_t iseg_edge_start[2] = {0, 0};
    auto alloc_seg = [&](seg_t &sg, uint64_t keys) {
        if (!keys) return;
        sg.bucket_start = bucket_cursor;
        sg.num_buckets = (keys + 4) / 5;
        bucket_cursor += sg.num_buckets;
    };
    for (uint32_t p = 0; p <= max_pid; p++) {
        alloc_seg(st->nseg[p * 2 + DIR_OUT], out_keys[p]);
        seg_edge_start[p * 2 + DIR_OUT] = edge_cursor; edge_cursor += out_edges[p];
        alloc_seg(st->nseg[p * 2 + DIR_IN], in_keys[p]);
        seg_edge_start[p * 2 + DIR_IN] = edge_cursor; edge_cursor += in_edges[p];
    }
    alloc_seg(st->iseg[DIR_IN], idx_in_keys);
    iseg_edge_start[DIR_IN] = edge_cursor; edge_cursor += idx_in_edges;
    alloc_seg(st->iseg[DIR_OUT], idx_out_keys);
    iseg_edge_start[DIR_OUT] = edge_cursor; edge_cursor += idx_out_edges;

    WK_LOG("[store] idx counts: %.1fs\n", now_s() - t0); t0 = now_s();
    st->nbuckets_main = bucket_cursor;
    st->nbuckets_ext = bucket_cursor / 4 + 1024;
    if (!st->vertices.alloc((st->nbuckets_main + st->nbuckets_ext) * ASSOC,
                            /*zero=*/true) ||
        !st->edges.alloc(edge_cursor, /*zero=*/false)) {
        WK_LOG("[store] allocation failed (%lu buckets, %lu edges)\n",
               (unsigned long)st->nbuckets_main, (unsigned long)edge_cursor);
        delete st;
        return nullptr;
    }
    std::atomic<uint64_t> ext_next(st->nbuckets_main);
    WK_LOG("[store] count+alloc: %.1fs (buckets=%lu edges=%lu)\n", now_s() - t0,
           (unsigned long)st->nbuckets_main, (unsigned long)edge_cursor);
    t0 = now_s();

    // 6. insert normal segments (parallel over (pid,dir) segments —
    //    static_gstore.hpp:64-161 semantics: one key per (vid,pid,dir) run,
    //    edge list = run values in sorted order)
#pragma omp parallel for schedule(dynamic)
    for (uint32_t w = 0; w < NP * 2; w++) {
        uint32_t p = w / 2;
        int dir = (int)(w & 1);
        bool out = dir == DIR_OUT;
        const seg_t &sg = st->nseg[w];
        if (!sg.num_buckets) continue;
        inserter ins{*st, sg, ext_next};
        uint64_t off = seg_edge_start[w];
        const std::vector<triple> &arr = out ? pso : pos;
        int64_t lo = out ? pso_lo[p] : pos_lo[p];
        int64_t hi = out ? pso_lo[p + 1] : pos_lo[p + 1];
        for (int64_t i = lo; i < hi;) {
            sid_t v = out ? arr[i].s : arr[i].o;
            int64_t j = i + 1;
            while (j < hi && (out ? arr[j].s : arr[j].o) == v) j++;
            if (!out && is_tpid(v)) { i = j; continue; }
            uint64_t slot = ins.insert(key_pack(v, p, (uint64_t)dir));
            st->vertices[slot].ptr = ptr_pack((uint64_t)(j - i), off);
            for (int64_t k = i; k < j; k++)
                st->edges[off++] = out ? arr[k].o : arr[k].s;
            i = j;
        }
    }

    WK_LOG("[store] insert normal: %.1fs\n", now_s() - t0); t0 = now_s();
    // 7. insert index segments (insert_idx, static_gstore.hpp:217-280;
    //    ours iterates pids ascending — deterministic where the reference's
    //    TBB iteration order was not; parity is set-level)
    {
        // IN: predicate-index (subjects) + type index
        if (st->iseg[DIR_IN].num_buckets) {
            inserter ins{*st, st->iseg[DIR_IN], ext_next};
            uint64_t off = iseg_edge_start[DIR_IN];
            for (uint32_t p = 2; p <= max_pid; p++) {
                if (out_keys[p]) {
                    uint64_t slot = ins.insert(key_pack(0, p, DIR_IN));
                    st->vertices[slot].ptr = ptr_pack(out_keys[p], off);
                    for (int64_t i = pso_lo[p]; i < pso_lo[p + 1];) {
                        int64_t j = i + 1;
                        while (j < pso_lo[p + 1] && pso[j].s == pso[i].s) j++;
                        st->edges[off++] = pso[i].s;
                        i = j;
                    }
                }
                if (type_members[p]) {
                    uint64_t slot = ins.insert(key_pack(0, p, DIR_IN));
                    st->vertices[slot].ptr = ptr_pack(type_members[p], off);
                    // members in subject order (type slice is (s,o)-sorted)
                    for (int64_t i = pso_lo[TYPE_ID]; i < pso_lo[TYPE_ID + 1]; i++)
                        if (pso[i].o == p) st->edges[off++] = pso[i].s;
                }
            }
        }
        // OUT: predicate-index (objects)
        if (st->iseg[DIR_OUT].num_buckets) {
            inserter ins{*st, st->iseg[DIR_OUT], ext_next};
            uint64_t off = iseg_edge_start[DIR_OUT];
            for (uint32_t p = 2; p <= max_pid; p++) {
                if (!in_keys[p]) continue;
                uint64_t slot = ins.insert(key_pack(0, p, DIR_OUT));
                st->vertices[slot].ptr = ptr_pack(in_keys[p], off);
                for (int64_t i = pos_lo[p]; i < pos_lo[p + 1];) {
                    int64_t j = i + 1;
                    while (j < pos_lo[p + 1] && pos[j].o == pos[i].o) j++;
                    if (!is_tpid(pos[i].o)) st->edges[off++] = pos[i].o;
                    i = j;
                }
            }
        }
    }
    WK_LOG("[store] insert index: %.1fs (ext used %lu)\n", now_s() - t0,
           (unsigned long)(ext_next.load() - st->nbuckets_main));
    // dense type side-index from the TYPE_ID pso slice (runs per subject)
    if (dense_vids && max_id >= (1u << NBITS_IDX) && max_pid < 0xFFFF) {
        st->type_base = 1u << NBITS_IDX;
        st->type_n = (uint64_t)max_id + 1 - st->type_base;
        if (!st->type_of.alloc(st->type_n, /*zero=*/true)) {
            st->type_n = 0;  // side index optional: probe fallback
        } else {
#pragma omp parallel for schedule(static)
            for (int64_t i = pso_lo[TYPE_ID]; i < pso_lo[TYPE_ID + 1]; i++) {
                bool first = (i == pso_lo[TYPE_ID]) || (pso[i - 1].s != pso[i].s);
                bool last = (i + 1 == pso_lo[TYPE_ID + 1]) || (pso[i + 1].s != pso[i].s);
                if (first && last) {
                    st->type_of[pso[i].s - st->type_base] = (uint16_t)pso[i].o;
                } else {
                    st->type_of[pso[i].s - st->type_base] = 0xFFFF;
                    st->type_multi = true;
                }
            }
        }
    }
    WK_LOG("[store] type index: %.1fs\n", now_s() - t0);
    t0 = now_s();
    // VERSATILE per-vertex predicate lists ([vid|PREDICATE_ID|dir],
    // static_gstore.hpp:282-374: OUT from pso runs INCLUDING type
    // triples; IN from pos runs skipping tpid objects) — dense CSR over
    // the vid range instead of hash keys (vids are dense; probe = 2
    // loads).  Lists are ascending-pid (built by ascending-p slices),
    // matching the sorted-edge-list invariant.  WK_VERSATILE=0 disables.
    {
        const char *vv = getenv("WK_VERSATILE");
        bool want = !(vv && !atoi(vv));
        if (want && dense_vids && max_id >= (1u << NBITS_IDX)) {
            st->vp_base = 1u << NBITS_IDX;
            st->vp_n = (uint64_t)max_id + 1 - st->vp_base;
            for (int dir = 0; dir < 2; dir++) {
                const bool out = dir == DIR_OUT;
                const std::vector<triple> &arr = out ? pso : pos;
                const std::vector<int64_t> &lo = out ? pso_lo : pos_lo;
                if (!st->vp_off[dir].alloc(st->vp_n + 1, /*zero=*/true)) {
                    st->vp_n = 0;  // VERSATILE optional: disable on OOM
                    st->vp_off[0].alloc(0, false);
                    st->vp_off[1].alloc(0, false);
                    break;
                }
                uint32_t *offp = st->vp_off[dir].data();
                // count distinct (vid, p) pairs per vid (shifted by 1
                // for the in-place exclusive scan)
#pragma omp parallel for schedule(dynamic)
                for (uint32_t p = 0; p <= max_pid; p++) {
                    for (int64_t i = lo[p]; i < lo[p + 1];) {
                        sid_t v = out ? arr[i].s : arr[i].o;
                        int64_t j = i + 1;
                        while (j < lo[p + 1] && (out ? arr[j].s : arr[j].o) == v)
                            j++;
                        if ((uint64_t)v >= st->vp_base && !(!out && is_tpid(v)))
                            __atomic_fetch_add(&offp[v - st->vp_base + 1], 1u,
                                               __ATOMIC_RELAXED);
                        i = j;
                    }
                }
                uint64_t total = 0;
                for (uint64_t i = 1; i <= st->vp_n; i++) {
                    total += offp[i];
                    if (total > 0xFFFFFFFFull) { total = 0; break; }
                    offp[i] = (uint32_t)total;
                }
                if (!total && st->vp_n) {  // u32 overflow: disable VERSATILE
                    st->vp_n = 0;
                    st->vp_off[0].alloc(0, false);
                    st->vp_off[1].alloc(0, false);
                    break;
                }
                if (!st->vp_edges[dir].alloc(total, /*zero=*/false)) {
                    st->vp_n = 0;
                    st->vp_off[0].alloc(0, false);
                    st->vp_off[1].alloc(0, false);
                    break;
                }
                std::vector<uint32_t> cursor(offp, offp + st->vp_n);
                // fill: sequential ascending p, parallel over runs (each
                // vid appears once per slice -> no cursor races)
                for (uint32_t p = 0; p <= max_pid; p++) {
#pragma omp parallel for schedule(static)
                    for (int64_t i = lo[p]; i < lo[p + 1]; i++) {
                        sid_t v = out ? arr[i].s : arr[i].o;
                        bool first = (i == lo[p]) ||
                                     (out ? arr[i - 1].s : arr[i - 1].o) != v;
                        if (!first) continue;
                        if ((uint64_t)v < st->vp_base || (!out && is_tpid(v)))
                            continue;
                        st->vp_edges[dir][cursor[v - st->vp_base]++] = p;
                    }
                }
            }
        }
    }
    WK_LOG("[store] versatile vp: %.1fs\n", now_s() - t0);
    t0 = now_s();
    // functional-predicate dense maps (keys == edges -> every key deg 1).
    // Budgeted: largest segments first until WK_FN_BUDGET_GB (default 24)
    // of host+HBM side-index space is spent.  WK_FN=0 disables.
    {
        const char *fv = getenv("WK_FN");
        bool want = !(fv && !atoi(fv));
        if (want && dense_vids && max_id >= (1u << NBITS_IDX)) {
            st->fn_base = 1u << NBITS_IDX;
            st->fn_n = (uint64_t)max_id + 1 - st->fn_base;
            st->fn.assign((size_t)NP * 2, {});
            const char *bv = getenv("WK_FN_BUDGET_GB");
            // default budget shrinks with the partition count: each
            // rank's maps still span the FULL vid range (mostly zeros
            // off-partition), so 8 ranks would cost 8x the host+HBM
            uint64_t gb = bv ? strtoull(bv, nullptr, 10)
                             : (uint64_t)(24 / (nsrv > 0 ? nsrv : 1));
            uint64_t budget = std::max<uint64_t>(gb, 2) << 30;
            std::vector<std::pair<uint64_t, uint32_t>> cand;  // (-edges, w)
            for (uint32_t p = 1; p <= max_pid; p++) {
                if (out_keys[p] && out_keys[p] == out_edges[p])
                    cand.push_back({out_edges[p], p * 2 + DIR_OUT});
                if (in_keys[p] && in_keys[p] == in_edges[p])
                    cand.push_back({in_edges[p], p * 2 + DIR_IN});
            }
            std::sort(cand.begin(), cand.end(),
                      [](auto &a, auto &b) { return a.first > b.first; });
            std::vector<uint32_t> picked;
            uint64_t spent = 0;
            const uint64_t npages = (st->fn_n + 63) / 64;
            for (auto &c : cand) {
                uint64_t bytes = npages * sizeof(fnpage_t) + c.first * 4;
                if (spent + bytes > budget) break;
                spent += bytes;
                picked.push_back(c.second);
            }
#pragma omp parallel for schedule(dynamic)
            for (size_t ci = 0; ci < picked.size(); ci++) {
                uint32_t w = picked[ci];
                uint32_t p = w / 2;
                bool out = (w & 1) == DIR_OUT;
                auto &m = st->fn[w];
                m.pages.assign(npages, fnpage_t{0, 0, 0});
                const std::vector<triple> &arr = out ? pso : pos;
                const std::vector<int64_t> &lo = out ? pso_lo : pos_lo;
                // keys ascend within the slice (deg==1: one triple per
                // key), so vals fill in rank order directly
                m.vals.reserve((size_t)(lo[p + 1] - lo[p]));
                for (int64_t i = lo[p]; i < lo[p + 1]; i++) {
                    sid_t v = out ? arr[i].s : arr[i].o;
                    if ((uint64_t)v < st->fn_base) continue;
                    uint64_t idx = v - st->fn_base;
                    m.pages[idx >> 6].bits |= 1ull << (idx & 63);
                    m.vals.push_back(out ? arr[i].o : arr[i].s);
                }
                uint32_t rank = 0;
                for (uint64_t g = 0; g < npages; g++) {
                    m.pages[g].rank = rank;
                    rank += (uint32_t)__builtin_popcountll(m.pages[g].bits);
                }
            }
            WK_LOG("[store] fn maps: %zu segments, %.1f GB (%.1fs)\n",
                   picked.size(), spent / 1e9, now_s() - t0);
        }
    }
    t0 = now_s();
    // rank-compressed CSR side index for NON-functional segments (the
    // fn-map structure generalized to any degree): pages + u64
    // {edge_off:40 | len:24} entries.  Probes drop from a 128-B bucket
    // walk to 24 B.  Budgeted largest-keys first after fn maps;
    // WK_CSR=0 disables.
    {
        const char *cv = getenv("WK_CSR");
        bool want = !(cv && !atoi(cv));
        if (want && dense_vids && max_id >= (1u << NBITS_IDX)) {
            if (!st->fn_n) {  // WK_FN=0: span bookkeeping still needed
                st->fn_base = 1u << NBITS_IDX;
                st->fn_n = (uint64_t)max_id + 1 - st->fn_base;
            }
            st->csr.assign((size_t)NP * 2, {});
            const char *bv = getenv("WK_CSR_BUDGET_GB");
            uint64_t gb = bv ? strtoull(bv, nullptr, 10)
                             : (uint64_t)(16 / (nsrv > 0 ? nsrv : 1));
            uint64_t budget = std::max<uint64_t>(gb, 2) << 30;
            const uint64_t npages = (st->fn_n + 63) / 64;
            std::vector<std::pair<uint64_t, uint32_t>> cand;  // (keys, w)
            auto has_fn = [&](uint32_t w) {
                return !st->fn.empty() && st->fn[w].present();
            };
            for (uint32_t p = 1; p <= max_pid; p++) {
                if (out_keys[p] && !has_fn(p * 2 + DIR_OUT))
                    cand.push_back({out_keys[p], p * 2 + DIR_OUT});
                if (in_keys[p] && !has_fn(p * 2 + DIR_IN))
                    cand.push_back({in_keys[p], p * 2 + DIR_IN});
            }
            std::sort(cand.begin(), cand.end(),
                      [](auto &a, auto &b) { return a.first > b.first; });
            std::vector<uint32_t> picked;
            uint64_t spent = 0;
            for (auto &c : cand) {
                uint64_t bytes = npages * sizeof(fnpage_t) + c.first * 8;
                if (spent + bytes > budget) break;
                spent += bytes;
                picked.push_back(c.second);
            }
#pragma omp parallel for schedule(dynamic)
            for (size_t ci = 0; ci < picked.size(); ci++) {
                uint32_t w = picked[ci];
                uint32_t p = w / 2;
                bool out = (w & 1) == DIR_OUT;
                auto &m = st->csr[w];
                m.pages.assign(npages, fnpage_t{0, 0, 0});
                const std::vector<triple> &arr = out ? pso : pos;
                const std::vector<int64_t> &lo = out ? pso_lo : pos_lo;
                m.entries.reserve((size_t)(out ? out_keys[p] : in_keys[p]));
                // edge offsets mirror the insert pass exactly: the
                // segment cursor advances one edge per kept triple
                uint64_t cur = seg_edge_start[w];
                bool ok = true;
                for (int64_t i = lo[p]; i < lo[p + 1] && ok;) {
                    sid_t v = out ? arr[i].s : arr[i].o;
                    int64_t j = i + 1;
                    while (j < lo[p + 1] && (out ? arr[j].s : arr[j].o) == v)
                        j++;
                    if (!out && is_tpid(v)) { i = j; continue; }
                    uint64_t len = (uint64_t)(j - i);
                    if ((uint64_t)v < st->fn_base || len >= (1ull << 24) ||
                        cur >= (1ull << 40)) {
                        ok = false;  // out of entry range: drop this map
                        break;
                    }
                    uint64_t idx = v - st->fn_base;
                    m.pages[idx >> 6].bits |= 1ull << (idx & 63);
                    m.entries.push_back((cur << 24) | len);
                    cur += len;
                    i = j;
                }
                if (!ok) {
                    m.pages.clear();
                    m.entries.clear();
                    continue;
                }
                uint32_t rank = 0;
                for (uint64_t g = 0; g < npages; g++) {
                    m.pages[g].rank = rank;
                    rank += (uint32_t)__builtin_popcountll(m.pages[g].bits);
                }
            }
            size_t nbuilt = 0;
            for (auto &m : st->csr) nbuilt += m.present();
            WK_LOG("[store] csr side index: %zu segments, %.1f GB (%.1fs)\n",
                   nbuilt, spent / 1e9, now_s() - t0);
        }
    }
    t0 = now_s();
    // per-type membership bitmaps from the TYPE_ID pso slice (exact:
    // includes multi-type vids).  Budgeted; WK_TBM=0 disables.
    {
        const char *tv = getenv("WK_TBM");
        bool want = !(tv && !atoi(tv));
        if (want && st->type_n) {
            const char *bv = getenv("WK_TBM_BUDGET_GB");
            uint64_t budget = (bv ? strtoull(bv, nullptr, 10) : 4) << 30;
            uint64_t words = (st->type_n + 63) / 64;
            uint64_t per = words * 8;
            st->tbm.assign((size_t)NP, {});
            // rank types by member count, biggest first
            std::vector<std::pair<uint64_t, uint32_t>> order;
            for (uint32_t t2 = 2; t2 < NP; t2++)
                if (type_members[t2]) order.push_back({type_members[t2], t2});
            std::sort(order.begin(), order.end(),
                      [](auto &a, auto &b) { return a.first > b.first; });
            uint64_t spent = 0;
            std::vector<uint32_t> picked;
            for (auto &pr : order) {
                if (spent + per > budget) break;
                spent += per;
                picked.push_back(pr.second);
            }
            for (uint32_t t2 : picked) st->tbm[t2].assign(words, 0);
            // one parallel sweep of the type slice (subject-disjoint
            // runs; same vid can repeat per type, but only within one
            // run -> races only on the same word for NEARBY subjects:
            // use atomic or bit OR; subjects within a 64-run may share a
            // word across threads, so use atomic OR
#pragma omp parallel for schedule(static)
            for (int64_t i = pso_lo[TYPE_ID]; i < pso_lo[TYPE_ID + 1]; i++) {
                uint32_t t2 = pso[i].o;
                if (t2 >= NP || st->tbm[t2].empty()) continue;
                uint64_t idx = (uint64_t)pso[i].s - st->type_base;
                if (idx >= st->type_n) continue;
                __atomic_fetch_or(&st->tbm[t2][idx >> 6],
                                  1ull << (idx & 63), __ATOMIC_RELAXED);
            }
            WK_LOG("[store] type bitmaps: %zu types, %.2f GB (%.1fs)\n",
                   picked.size(), spent / 1e9, now_s() - t0);
        }
    }
    st->ext_used = ext_next.load() - st->nbuckets_main;
    return st;
}

// C-ABI entry: allocation failure surfaces as nullptr, never as an
// exception crossing the boundary (the vector/sort scaffolding inside
// the build can throw bad_alloc on hosts smaller than the dataset; the
// pod_array side indexes already degrade in place).  The partially
// built store is abandoned on this path — the process is at the OOM
// edge and the caller's contract is "nullptr = build failed".
extern "C" wk_store_t *wk_store_build(const sid_t *spo, int64_t ntriples,
                                      int32_t sid, int32_t nsrv) {
    try {
        return store_build_impl(spo, ntriples, sid, nsrv);
    } catch (const std::bad_alloc &) {
        return nullptr;
    }
}

extern "C" void wk_store_free(wk_store_t *st) { delete st; }

extern "C" const sid_t *wk_store_get_triples(const wk_store_t *st, sid_t vid,
                                             sid_t pid, int32_t dir, uint64_t *sz) {
    return store_get(*st, vid, pid, dir, sz);
}

extern "C" const sid_t *wk_store_get_index(const wk_store_t *st, sid_t pid,
                                           int32_t dir, uint64_t *sz) {
    return store_get(*st, 0, pid, dir, sz);
}

// per-(pid,dir) key/edge counts — the planner's cost-model inputs
// (the reference's type-centric statistics, core/optimizer/stats.hpp)
extern "C" int32_t wk_store_seg_stats(const wk_store_t *st, uint32_t pid,
                                      int32_t dir, uint64_t *keys,
                                      uint64_t *edges) {
    if (!st || dir < 0 || dir > 1) return -1;
    size_t w = (size_t)pid * 2 + dir;
    if (w >= st->seg_keys.size()) {
        if (keys) *keys = 0;
        if (edges) *edges = 0;
        return 0;
    }
    if (keys) *keys = st->seg_keys[w];
    if (edges) *edges = st->seg_edges[w];
    return 0;
}

// memory-usage report (the reference's GStore::print_mem_usage,
// core/store/gstore.hpp:1062-1103): bytes per region, via out params.
extern "C" int32_t wk_store_mem_usage(const wk_store_t *st,
                                      uint64_t *slots_bytes,
                                      uint64_t *edges_bytes,
                                      uint64_t *side_index_bytes) {
    if (!st) return -1;
    if (slots_bytes) *slots_bytes = st->vertices.size() * sizeof(vertex_t);
    if (edges_bytes) *edges_bytes = st->edges.size() * sizeof(sid_t);
    uint64_t side = st->type_n * 2;  // type_of
    for (int d = 0; d < 2; d++)
        side += st->vp_off[d].size() * 4 + st->vp_edges[d].size() * 4;
    for (auto &m : st->fn)
        side += m.pages.size() * sizeof(fnpage_t) + m.vals.size() * 4;
    for (auto &m : st->csr)
        side += m.pages.size() * sizeof(fnpage_t) + m.entries.size() * 8;
    for (auto &b : st->tbm) side += b.size() * 8;
    if (side_index_bytes) *side_index_bytes = side;
    return 0;
}

extern "C" uint64_t wk_store_num_slots(const wk_store_t *st) {
    return st->vertices.size();
}
extern "C" uint64_t wk_store_num_edges(const wk_store_t *st) {
    return st->edges.size();
}
// Full-store integrity scan — the reference's `gsck` console command
// (GChecker::gstore_check, core/store/gchecker.hpp:364-392): every
// slot's edge extent in bounds; every predicate-index subject/object has
// the matching normal key; every type-index member lists that type.
// Returns 0 if consistent, else the number of violations.
extern "C" uint64_t wk_store_check(const wk_store_t *st) {
    std::atomic<uint64_t> bad(0);
    const uint64_t nslots = st->vertices.size();
    const uint64_t nedges = st->edges.size();
#pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)nslots; i++) {
        if (i % ASSOC == ASSOC - 1) continue;  // chain slot
        const vertex_t &v = st->vertices[i];
        if (v.key == KEY_EMPTY) continue;
        uint64_t off = ptr_off(v.ptr), sz = ptr_size(v.ptr);
        if (off + sz > nedges) { bad++; continue; }
        // normal-key edge lists must be ascending (loader sort invariant
        // the GPU filters rely on)
        if (key_vid(v.key) != 0) {
            for (uint64_t k = 1; k < sz; k++)
                if (st->edges[off + k - 1] >= st->edges[off + k]) { bad++; break; }
        }
    }
    // index <-> normal cross-check
    for (uint32_t pid = 2; pid <= st->max_pid; pid++) {
        for (int d = 0; d < 2; d++) {
            uint64_t sz = 0;
            const sid_t *lst = store_get(*st, 0, pid, d, &sz);
            if (!lst) continue;
#pragma omp parallel for schedule(static)
            for (int64_t k = 0; k < (int64_t)sz; k++) {
                sid_t v = lst[k];
                uint64_t esz = 0;
                if (is_tpid(v)) { bad++; continue; }
                // [0|pid|IN] lists subjects (OUT keys); [0|pid|OUT] objects
                // (IN keys); [0|tid|IN] lists type members ([v|TYPE|OUT])
                const sid_t *e1 = store_get(*st, v, pid, d == DIR_IN ? DIR_OUT : DIR_IN, &esz);
                if (e1 && esz) continue;
                // maybe pid is a type id: member must have [v|TYPE|OUT]
                // containing pid
                const sid_t *t = store_get(*st, v, TYPE_ID, DIR_OUT, &esz);
                bool ok = false;
                if (t)
                    for (uint64_t j = 0; j < esz; j++)
                        if (t[j] == pid) { ok = true; break; }
                if (!ok) bad++;
            }
        }
    }
    return bad.load();
}

extern "C" uint64_t wk_store_checksum(const wk_store_t *st) {
    // Content fingerprint: per occupied slot, FNV-1a over (key, edge
    // list), combined with a commutative wrapping sum.  Bucket/chain
    // placement is left OUT on purpose — parallel ext-bucket allocation
    // makes the physical layout run-to-run nondeterministic, but two
    // stores with the same logical triples must fingerprint equal.
    std::atomic<uint64_t> total(0);
    const int64_t nslots = (int64_t)st->vertices.size();
#pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < nslots; i++) {
        if (i % ASSOC == ASSOC - 1) continue;  // chain slot
        const vertex_t &v = st->vertices[i];
        if (v.key == KEY_EMPTY) continue;
        uint64_t h = 1469598103934665603ull;
        auto mix = [&h](uint64_t x) {
            for (int b = 0; b < 8; b++) { h ^= (x >> (8 * b)) & 0xff; h *= 1099511628211ull; }
        };
        mix(v.key);
        uint64_t off = ptr_off(v.ptr), sz = ptr_size(v.ptr);
        mix(sz);
        for (uint64_t k = 0; k < sz && off + k < st->edges.size(); k++)
            mix(st->edges[off + k]);
        total.fetch_add(h, std::memory_order_relaxed);
    }
    return total.load();
}
