/*
 * wk_store.h — host-side store object shared by the builder (store.cpp)
 * and the GPU engine (gpu_engine.hip).  Layout semantics: DESIGN.md §2,
 * restating core/store/gstore.hpp + static_gstore.hpp.
 */
#pragma once
#include "wk_types.h"
#include <vector>
#include <cstdlib>
#include <cstring>
#include <algorithm>

// flat POD array with parallel zeroing (std::vector's value-init is
// single-threaded and page-fault-bound: ~6 s for the 1 GB slot array)
template <class T>
struct pod_array {
    T *ptr = nullptr;
    size_t n = 0;
    pod_array() = default;
    pod_array(const pod_array &) = delete;
    pod_array &operator=(const pod_array &) = delete;
    ~pod_array() { free(ptr); }
    // returns false on allocation failure (n stays 0; wk_store_build
    // propagates it as a NULL store instead of faulting in the memset)
    bool alloc(size_t count, bool zero) {
        free(ptr);
        ptr = (T *)malloc(count * sizeof(T) + 16);
        n = 0;
        if (!ptr) return false;
        n = count;
        if (zero) {
#pragma omp parallel for schedule(static)
            for (long long i = 0; i < (long long)count; i += (1 << 20)) {
                size_t len = std::min<size_t>(1 << 20, count - (size_t)i);
                memset(ptr + i, 0, len * sizeof(T));
            }
        }
        return true;
    }
    T *data() { return ptr; }
    const T *data() const { return ptr; }
    size_t size() const { return n; }
    T *begin() { return ptr; }
    T *end() { return ptr + n; }
    const T *begin() const { return ptr; }
    const T *end() const { return ptr + n; }
    T &operator[](size_t i) { return ptr[i]; }
    const T &operator[](size_t i) const { return ptr[i]; }
};

struct wk_store {
    int32_t sid = 0, nsrv = 1;
    uint32_t max_pid = 0;  // largest predicate/type id present

    // cluster-hash slot array: (main + ext buckets) * 8 slots
    pod_array<wk::vertex_t> vertices;
    // all edge lists, contiguous per segment
    pod_array<wk::sid_t> edges;
    // normal segments indexed [pid*2 + dir], size (max_pid+1)*2
    std::vector<wk::seg_t> nseg;

    // dense single-type side index: type_of[vid - type_base] = the type id
    // of vid (0 = untyped, 0xFFFF = multiple types -> probe fallback).
    // Accelerates `?X rdf:type T` filters to one 2-byte read; derived
    // from the [vid|TYPE_ID|OUT] lists, bit-identical results.
    pod_array<uint16_t> type_of;
    uint64_t type_base = 0, type_n = 0;
    bool type_multi = false;  // any entity with >1 type (0xFFFF fallbacks)
    // index segments [dir]: keys [0|pid|dir] / [0|tid|IN]
    wk::seg_t iseg[2];

    // VERSATILE per-vertex predicate lists ([vid|PREDICATE_ID|dir],
    // static_gstore.hpp:282-374 semantics: OUT includes rdf:type, IN
    // skips type triples) stored as a dense CSR over the vid range —
    // vids are dense, so 2 loads replace the cluster-hash walk
    // (same deliberate densification as type_of).
    pod_array<uint32_t> vp_off[2];   // size vp_n+1 (CSR offsets)
    pod_array<wk::sid_t> vp_edges[2];
    uint64_t vp_base = 0, vp_n = 0;

    // functional-predicate maps: for a (pid,dir) segment where EVERY
    // key has degree exactly 1 (LUBM: memberOf, ugDegreeFrom, worksFor,
    // advisor, name, ... — detected as keys==edges), a rank-compressed
    // map (presence-bitmap pages + packed values, wk_types.h fnpage_t)
    // replaces the 148-byte cluster-hash probe with a 16-B page load +
    // 4-B value load (0 = vid has no such edge; 0 is never a valid id).
    // Results identical; an HBM-capacity-funded densification like
    // type_of.  Indexed [pid*2+dir]; empty = absent.
    struct fnmap {
        std::vector<wk::fnpage_t> pages;  // (fn_n+63)/64
        std::vector<wk::sid_t> vals;      // nkeys, rank-indexed
        bool present() const { return !pages.empty(); }
    };
    std::vector<fnmap> fn;
    uint64_t fn_base = 0, fn_n = 0;

    // rank-compressed CSR side index for NON-functional segments (the
    // fn-map idea generalized): pages as above, entries u64
    // {edge_off:40 | len:24} rank-indexed.  A k2u probe becomes a 16-B
    // page load + 8-B entry load instead of a 128-B bucket walk (the
    // cluster-hash stays the keyed layout of record; this is another
    // HBM-capacity-funded densification like type_of/fn).  Indexed
    // [pid*2+dir]; empty = absent.
    struct csrmap {
        std::vector<wk::fnpage_t> pages;
        std::vector<uint64_t> entries;
        bool present() const { return !pages.empty(); }
    };
    std::vector<csrmap> csr;

    // host-side lookup (mirrors the device fn_lookup)
    wk::sid_t fn_get(size_t w, wk::sid_t v) const {
        if (w >= fn.size() || !fn[w].present()) return 0;
        return wk::fn_lookup(fn[w].pages.data(), fn[w].vals.data(),
                             fn_base, fn_n, v);
    }

    // per-(pid,dir) key/edge counts — the planner's cost-model inputs
    // (the reference's type-centric stats, core/optimizer/stats.hpp)
    std::vector<uint64_t> seg_keys, seg_edges;  // [pid*2+dir]

    // per-type membership bitmaps: tbm[tid] bit (vid - vp_base) set iff
    // (vid, TYPE_ID, tid) exists.  1 bit/vid (~22 MB per type at
    // LUBM-2560) keeps the whole bitmap LLC-resident, so `?x rdf:type T`
    // filters read cache lines instead of HBM-random 2-byte gathers —
    // and the semantics are EXACT (multi-type vids included; no 0xFFFF
    // fallback).  Indexed by type id; empty = absent.
    std::vector<std::vector<uint64_t>> tbm;

    uint64_t nbuckets_main = 0, nbuckets_ext = 0, ext_used = 0;

    const wk::seg_t *seg_of(uint64_t vid, uint64_t pid, int dir) const {
        if (vid == 0) return &iseg[dir];
        if (pid > max_pid) return nullptr;
        const wk::seg_t *s = &nseg[pid * 2 + dir];
        return s->num_buckets ? s : nullptr;
    }
};

// host probe — mirrors GStore::get_vertex_local (gstore.hpp:341-361)
namespace wk {
inline const sid_t *store_get(const wk_store &st, uint64_t vid, uint64_t pid,
                              int dir, uint64_t *sz) {
    *sz = 0;
    // [vid|PREDICATE_ID|dir] = the vertex's predicate list (VERSATILE,
    // gstore get_triples surface) — served from the dense CSR
    if (pid == PREDICATE_ID && vid != 0) {
        uint64_t idx = vid - st.vp_base;
        if (idx >= st.vp_n) return nullptr;
        uint64_t lo = st.vp_off[dir][idx], hi = st.vp_off[dir][idx + 1];
        *sz = hi - lo;
        return *sz ? st.vp_edges[dir].data() + lo : nullptr;
    }
    const seg_t *seg = st.seg_of(vid, pid, dir);
    if (!seg || seg->num_buckets == 0) return nullptr;
    uint64_t key = key_pack(vid, pid, (uint64_t)dir);
    uint64_t bucket = seg->bucket_start + hash_u64(key) % seg->num_buckets;
    while (true) {
        const vertex_t *b = &st.vertices[bucket * ASSOC];
        for (int i = 0; i < ASSOC - 1; i++)
            if (b[i].key == key) {
                *sz = ptr_size(b[i].ptr);
                return st.edges.data() + ptr_off(b[i].ptr);
            }
        if (b[ASSOC - 1].key == KEY_EMPTY) return nullptr;
        bucket = key_vid(b[ASSOC - 1].key);  // chain (gstore.hpp:826)
    }
}
}  // namespace wk
