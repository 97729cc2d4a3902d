/*
 * ORACLE — TEST INFRASTRUCTURE ONLY.
 *
 * Dependency-free C++17 CPU restatement of the reference engine's hot
 * path, used exclusively as the parity checker and as bench.py's
 * `cpu_baseline` leg.  Only `tests/`, `__graft_entry__.smoke()` and
 * `bench.py` may link/load/call this library.  The product path
 * (wukong_amd/) must never import or fall back to it.
 *
 * What it restates (reference = SJTU-IPADS/wukong):
 *  - loader partition/sort/dedup:  core/loader/base_loader.hpp:302-373
 *  - store semantics (edge lists per [vid|pid|dir] key, predicate/type
 *    indexes):                     core/store/gstore.hpp:55-120,858-888,
 *                                  core/store/static_gstore.hpp:64-280
 *  - key/ptr bit layouts + hash:   core/store/vertex.hpp:34-151,
 *                                  utils/math.hpp:51-66
 *  - operators:                    core/engine/sparql.hpp:80-549
 *      index_to_unknown :194-231, const_to_unknown :238-285,
 *      known_to_unknown :295-407, known_to_known :416-476,
 *      known_to_const :484-549, const_to_known :138-186,
 *      index_to_known :80-135
 *  - dispatch:                     core/engine/sparql.hpp:1016-1058
 *  - mt_factor index slicing:      core/engine/sparql.hpp:210-221
 *  - final (DISTINCT/OFFSET/LIMIT/projection): sparql.hpp:1424-1551
 *
 * Parity pinning: the reference does not compile in this
 * container (boost/TBB/MPI absent — SURVEY.md §8c), so the oracle is
 * pinned by (a) golden hash/bit-layout vectors dumped from the
 * reference's OWN standalone headers (oracle/ref_dump.cpp →
 * tests/golden/hash_golden.csv), and (b) an algorithmically independent
 * brute-force BGP evaluator in brute.cpp (hash-join over the raw triple
 * list, no store, no exploration) that must agree on every query.
 * Count-level parity vs the reference's published LUBM-2560 numbers is
 * PARTIAL: real LUBM data is not generatable here (Java generator).
 *
 * The store here is deliberately a *different* data structure from the
 * product's cluster-hash (std::unordered_map keyed by the packed 64-bit
 * ikey) — same key encoding and edge-list semantics, independent layout.
 */
#include <cstdint>
#include <cstring>
#include <cstdlib>
#include <vector>
#include <unordered_map>
#include <unordered_set>
#include <algorithm>
#include <omp.h>
#include "ok_internal.h"

namespace ok {

// TomasWang hash — utils/math.hpp:58-66
static inline uint64_t hash_u64(uint64_t key) {
    key = (~key) + (key << 21);
    key = key ^ (key >> 24);
    key = (key + (key << 3)) + (key << 8);
    key = key ^ (key >> 14);
    key = (key + (key << 2)) + (key << 4);
    key = key ^ (key >> 28);
    key = key + (key << 31);
    return key;
}

// parallel merge sort (chunk std::sort + pairwise inplace_merge): same
// comparator, same total order as a single std::sort
template <class T, class Cmp>
static void par_sort(std::vector<T> &v, Cmp cmp) {
    const size_t n = v.size();
    int chunks = omp_get_max_threads();
    if (chunks <= 1 || n < (1u << 16)) {
        std::sort(v.begin(), v.end(), cmp);
        return;
    }
    std::vector<size_t> bnd(chunks + 1);
    for (int i = 0; i <= chunks; i++) bnd[i] = n * (size_t)i / chunks;
#pragma omp parallel for schedule(dynamic, 1)
    for (int i = 0; i < chunks; i++)
        std::sort(v.begin() + bnd[i], v.begin() + bnd[i + 1], cmp);
    for (int step = 1; step < chunks; step *= 2) {
#pragma omp parallel for schedule(dynamic, 1)
        for (int i = 0; i < chunks; i += 2 * step) {
            int mid = i + step, end = std::min(i + 2 * step, chunks);
            if (mid < end)
                std::inplace_merge(v.begin() + bnd[i], v.begin() + bnd[mid],
                                   v.begin() + bnd[end], cmp);
        }
    }
}

// ---- build (loader + static store semantics) -------------------------
// The build phase is OpenMP-parallel (partition, sort, fills) purely for
// wall-clock — bench.py's cpu_baseline leg builds this store at full
// LUBM-2560 scale.  Query semantics and all stored orders are identical
// to the sequential form.
static ctx *build(const sid_t *spo, int64_t n, int sid, int nsrv) {
    ctx *c = new ctx();
    c->sid = sid; c->nsrv = nsrv;

    std::vector<triple> pso, pos;
    {
        int nt = omp_get_max_threads();
        std::vector<std::vector<triple>> lso(nt), los(nt);
#pragma omp parallel num_threads(nt)
        {
            int t = omp_get_thread_num();
            lso[t].reserve((size_t)(n / nt) + 64);
            los[t].reserve((size_t)(n / nt) + 64);
#pragma omp for schedule(static)
            for (int64_t i = 0; i < n; i++) {
                triple tr{spo[3 * i], spo[3 * i + 1], spo[3 * i + 2]};
                // partition — base_loader.hpp:344-352
                if ((int)(tr.s % (sid_t)nsrv) == sid) lso[t].push_back(tr);
                if ((int)(tr.o % (sid_t)nsrv) == sid) los[t].push_back(tr);
            }
        }
        size_t tso = 0, tos = 0;
        for (int t = 0; t < nt; t++) { tso += lso[t].size(); tos += los[t].size(); }
        pso.resize(tso); pos.resize(tos);
        size_t oso = 0, oos = 0;
        for (int t = 0; t < nt; t++) {
            memcpy(pso.data() + oso, lso[t].data(), lso[t].size() * sizeof(triple));
            memcpy(pos.data() + oos, los[t].data(), los[t].size() * sizeof(triple));
            oso += lso[t].size(); oos += los[t].size();
        }
    }
    // sort + dedup — base_loader.hpp:367-377 (parallel merge sort: same
    // comparator, same total order as std::sort)
    par_sort(pso, [](const triple &a, const triple &b) {
        return a.p != b.p ? a.p < b.p : (a.s != b.s ? a.s < b.s : a.o < b.o);
    });
    par_sort(pos, [](const triple &a, const triple &b) {
        return a.p != b.p ? a.p < b.p : (a.o != b.o ? a.o < b.o : a.s < b.s);
    });
    auto eq = [](const triple &a, const triple &b) {
        return a.s == b.s && a.p == b.p && a.o == b.o;
    };
    pso.erase(std::unique(pso.begin(), pso.end(), eq), pso.end());
    pos.erase(std::unique(pos.begin(), pos.end(), eq), pos.end());

    // ---- slab geometry -------------------------------------------------
    // OUT edge lists are exactly pso's object column in order (one edge
    // per triple), so the OUT slab is [0, |pso|) and a run starting at i
    // has offset i.  The IN slab skips tpid-object runs; tpid objects
    // sort FIRST within each predicate slice (ids < 2^17), so each
    // slice's kept region is contiguous and offsets stay derivable.
    sid_t max_pid = 1;
#pragma omp parallel for reduction(max : max_pid) schedule(static)
    for (int64_t i = 0; i < (int64_t)pos.size(); i++)
        max_pid = std::max(max_pid, pos[i].p);
    const size_t NPD = (size_t)max_pid + 1;
    std::vector<size_t> slice_lo(NPD + 1, 0), skip(NPD, 0), seg(NPD + 1, 0);
    {
        size_t i = 0;
        for (size_t p = 0; p < NPD; p++) {
            slice_lo[p] = i;
            while (i < pos.size() && pos[i].p == (sid_t)p) i++;
        }
        slice_lo[NPD] = pos.size();
        for (size_t p = 0; p < NPD; p++) {
            size_t lo = slice_lo[p], hi = slice_lo[p + 1];
            size_t s2 = lo;
            while (s2 < hi && is_tpid(pos[s2].o)) s2++;
            skip[p] = s2 - lo;
            seg[p + 1] = seg[p] + (hi - s2);
        }
    }
    const size_t out_n = pso.size(), in_n = seg[NPD];
    c->edges.resize(out_n + in_n);
#pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)out_n; i++) c->edges[i] = pso[i].o;
#pragma omp parallel for schedule(dynamic)
    for (size_t p = 0; p < NPD; p++)
        for (size_t i = slice_lo[p] + skip[p]; i < slice_lo[p + 1]; i++)
            c->edges[out_n + seg[p] + (i - slice_lo[p] - skip[p])] = pos[i].s;

    // ---- kv inserts, parallel over shards ------------------------------
    // Each shard task scans the run starts and keeps only its keys; a
    // run's offset is derivable from its start index (above), so shards
    // are fully independent.  Same (key -> off,len) map as the
    // sequential form.
    auto insert_runs = [&](bool out) {
        const std::vector<triple> &arr = out ? pso : pos;
        const size_t base = out ? 0 : out_n;
#pragma omp parallel for schedule(dynamic, 1)
        for (int sh = 0; sh < ctx::NSHARD; sh++) {
            auto &m = c->kv[sh];
            for (size_t i = 0; i < arr.size();) {
                sid_t v = out ? arr[i].s : arr[i].o;
                size_t j = i + 1;
                while (j < arr.size() && arr[j].p == arr[i].p &&
                       (out ? arr[j].s : arr[j].o) == v)
                    j++;
                if (!out && is_tpid(v)) { i = j; continue; }
                uint64_t key = key_pack(v, arr[i].p, (uint64_t)(out ? DIR_OUT : DIR_IN));
                if (ctx::shard_of(key) == sh) {
                    size_t p = arr[i].p;
                    size_t off = out ? i
                                     : base + seg[p] + (i - slice_lo[p] - skip[p]);
                    m[key] = {off, j - i};
                }
                i = j;
            }
        }
    };
    insert_runs(true);   // OUT keys [s|p|OUT] — static_gstore.hpp:95-117
    insert_runs(false);  // IN keys [o|p|IN], tpid objects skipped — :125-152

    // ---- index keys + VERSATILE ----------------------------------------
    // pidx/tidx (gstore.hpp:858-888) are tiny (one list per predicate /
    // type): sequential run scan, no edge copies
    auto put = [&](uint64_t vid, uint64_t pid, int dir, const std::vector<sid_t> &vals) {
        uint64_t key = key_pack(vid, pid, (uint64_t)dir);
        c->kv[ctx::shard_of(key)][key] = {c->edges.size(), vals.size()};
        c->edges.insert(c->edges.end(), vals.begin(), vals.end());
    };
    std::unordered_map<sid_t, std::vector<sid_t>> pidx_in, pidx_out, tidx;
    std::vector<std::pair<sid_t, sid_t>> vpo, vpi;  // (vid, pid) per run
    vpo.reserve(out_n / 2); vpi.reserve(in_n / 2);
    for (size_t i = 0; i < pso.size();) {
        size_t j = i + 1;
        while (j < pso.size() && pso[j].p == pso[i].p && pso[j].s == pso[i].s) j++;
        vpo.push_back({pso[i].s, pso[i].p});
        if (pso[i].p == TYPE_ID)
            for (size_t k = i; k < j; k++) tidx[pso[k].o].push_back(pso[i].s);
        else if (pso[i].p != PREDICATE_ID)
            pidx_in[pso[i].p].push_back(pso[i].s);
        i = j;
    }
    for (size_t i = 0; i < pos.size();) {
        size_t j = i + 1;
        while (j < pos.size() && pos[j].p == pos[i].p && pos[j].o == pos[i].o) j++;
        if (!is_tpid(pos[i].o)) {
            vpi.push_back({pos[i].o, pos[i].p});
            if (pos[i].p != PREDICATE_ID && pos[i].p != TYPE_ID)
                pidx_out[pos[i].p].push_back(pos[i].o);
        }
        i = j;
    }
    // [0|pid|IN] = subjects, [0|pid|OUT] = objects, [0|tid|IN] = members
    for (auto &kvp : pidx_in) put(0, kvp.first, DIR_IN, kvp.second);
    for (auto &kvp : pidx_out) put(0, kvp.first, DIR_OUT, kvp.second);
    for (auto &kvp : tidx) put(0, kvp.first, DIR_IN, kvp.second);
    // VERSATILE per-vertex predicate lists [vid|PREDICATE_ID|dir] —
    // insert_vp, static_gstore.hpp:282-374: OUT from pso runs INCLUDING
    // type triples; IN from pos runs skipping tpid objects.  One
    // (vid, pid) pair per run; (vid, pid)-sorting yields each vid's
    // predicate list ascending — the same lists the per-vid-map + sort
    // form produced.  One edge per pair, so a vid run starting at k has
    // offset vp_base + k (shard-parallel inserts as above).
    for (int dir = 0; dir < 2; dir++) {
        auto &vp = dir == DIR_OUT ? vpo : vpi;
        par_sort(vp, [](const std::pair<sid_t, sid_t> &a,
                        const std::pair<sid_t, sid_t> &b) { return a < b; });
        const size_t vp_base = c->edges.size();
        c->edges.resize(vp_base + vp.size());
#pragma omp parallel for schedule(static)
        for (int64_t k = 0; k < (int64_t)vp.size(); k++)
            c->edges[vp_base + k] = vp[k].second;
#pragma omp parallel for schedule(dynamic, 1)
        for (int sh = 0; sh < ctx::NSHARD; sh++) {
            auto &m = c->kv[sh];
            for (size_t i = 0; i < vp.size();) {
                size_t j = i + 1;
                while (j < vp.size() && vp[j].first == vp[i].first) j++;
                uint64_t key =
                    key_pack(vp[i].first, PREDICATE_ID, (uint64_t)dir);
                if (ctx::shard_of(key) == sh) m[key] = {vp_base + i, j - i};
                i = j;
            }
        }
    }
    c->triples = std::move(pso);  // brute.cpp uses the pso-side dedup'd set
    return c;
}

// ---- engine ----------------------------------------------------------
struct pattern { ssid_t s, p, o; int dir; };

struct query {
    std::vector<pattern> pats;
    int nvars = 0;
    std::vector<int> v2c;        // query.hpp:352-374
    std::vector<sid_t> table;    // row-major
    int col_num = 0;
    int step = 0;
    int mt_tid = 0, mt_factor = 1;  // sparql.hpp:210-221

    // OPTIONAL group state (optional_matched_rows + optional_new_vars,
    // query.hpp:722-813): rows are kept; unmatched rows carry BLANK_ID
    // in the columns first bound inside the group.
    int opt_mode = 0;
    uint32_t opt_mask = 0;           // columns to blank on mismatch
    std::vector<uint8_t> matched;

    int var2col(ssid_t v) const { return v < 0 ? v2c[-(v + 1)] : -1; }
    int64_t nrows() const { return col_num ? (int64_t)table.size() / col_num : 0; }

    void blank_row(int64_t i) {      // correct_optional_result, query.hpp:805
        for (int c = 0; c < col_num; c++)
            if ((opt_mask >> c) & 1)
                table[i * col_num + c] = 0xFFFFFFFFu;  // BLANK_ID
    }
};
static const sid_t OBLANK = 0xFFFFFFFFu;

// one pattern — dispatch per sparql.hpp:1016-1058
static void exec_pattern(const ctx &c, query &q) {
    const pattern pat = q.pats[q.step];
    const ssid_t s = pat.s, p = pat.p, o = pat.o;
    const int d = pat.dir;
    uint64_t sz = 0;

    // index_to_unknown — sparql.hpp:194-231 (+ mt slicing :210-221)
    if (q.step == 0 && s >= 0 && is_tpid(s)) {
        const sid_t *edges = c.get(0, (uint64_t)s, d, &sz);
        std::vector<sid_t> out;
        uint64_t start = (uint64_t)(q.mt_tid % q.mt_factor);
        uint64_t len = sz / (uint64_t)q.mt_factor;
        for (uint64_t k = start * len; k < (start + 1) * len; k++) out.push_back(edges[k]);
        if (start == (uint64_t)q.mt_factor - 1)
            for (uint64_t k = (start + 1) * len; k < sz; k++) out.push_back(edges[k]);
        q.table.swap(out);
        q.col_num = 1;
        q.v2c[-(o + 1)] = 0;
        q.step++;
        return;
    }

    // VERSATILE ops (predicate variable) — sparql.hpp:556-744: read the
    // vertex's predicate list under [vid|PREDICATE_ID|dir], then probe
    // each (vid, p, dir); *_unknown_unknown appends (p, y) per edge,
    // *_unknown_const appends one row per matching p.
    if (p < 0) {
        std::vector<sid_t> out;
        auto run_vid = [&](sid_t vid, const sid_t *row, int rowlen) {
            uint64_t npids = 0;
            const sid_t *pids = c.get(vid, PREDICATE_ID, d, &npids);
            for (uint64_t pi = 0; pi < npids; pi++) {
                uint64_t esz = 0;
                const sid_t *vids2 = c.get(vid, pids[pi], d, &esz);
                if (o >= 0) {
                    for (uint64_t k = 0; k < esz; k++)
                        if (vids2[k] == (sid_t)o) {
                            out.insert(out.end(), row, row + rowlen);
                            out.push_back(pids[pi]);
                            break;
                        }
                } else {
                    for (uint64_t k = 0; k < esz; k++) {
                        out.insert(out.end(), row, row + rowlen);
                        out.push_back(pids[pi]);
                        out.push_back(vids2[k]);
                    }
                }
            }
        };
        if (s >= 0) {  // const_unknown_* (first pattern, sparql.hpp:719)
            run_vid((sid_t)s, nullptr, 0);
        } else {
            int col = q.var2col(s);
            int64_t R = q.nrows();
            for (int64_t i = 0; i < R; i++)
                run_vid(q.table[i * q.col_num + col],
                        q.table.data() + i * q.col_num, q.col_num);
        }
        q.table.swap(out);
        q.v2c[-(p + 1)] = q.col_num;
        // NOTE: when o is a var this REBINDS it unconditionally — the
        // known_unknown_known shape has no operator in the reference
        // (sparql.hpp:556-744) and the engine rejects it (WK_ERR_PLAN);
        // the planner never emits a vu after its object var binds, so
        // this branch only ever sees a fresh o var.
        if (o < 0) q.v2c[-(o + 1)] = q.col_num + 1;
        q.col_num += (o >= 0) ? 1 : 2;
        q.step++;
        return;
    }

    if (s >= 0) {
        const sid_t *vids = c.get((uint64_t)s, (uint64_t)p, d, &sz);
        if (o < 0 && q.var2col(o) < 0) {
            // const_to_unknown — sparql.hpp:250-263 (first pattern only)
            std::vector<sid_t> out(vids, vids + sz);
            q.table.swap(out);
            q.v2c[-(o + 1)] = q.col_num;
            q.col_num += 1;
        } else {
            // const_to_known — sparql.hpp:138-186 (+ OPTIONAL branch
            // :160-170: keep rows, blank + clear flag on mismatch)
            std::unordered_set<sid_t> set(vids, vids + sz);
            int col = q.var2col(o);
            int64_t R = q.nrows();
            if (q.opt_mode) {
                for (int64_t i = 0; i < R; i++) {
                    if (!set.count(q.table[i * q.col_num + col])) {
                        if (q.matched[i]) q.blank_row(i);
                        q.matched[i] = 0;
                    }
                }
            } else {
                std::vector<sid_t> out;
                for (int64_t i = 0; i < R; i++)
                    if (set.count(q.table[i * q.col_num + col]))
                        out.insert(out.end(), q.table.begin() + i * q.col_num,
                                   q.table.begin() + (i + 1) * q.col_num);
                q.table.swap(out);
            }
        }
        q.step++;
        return;
    }

    int col = q.var2col(s);
    int64_t R = q.nrows();
    const int ostat = (o >= 0) ? 2 : (q.var2col(o) >= 0 ? 1 : 0);

    // consecutive-dup memo — sparql.hpp:322-345
    sid_t cached = 0xFFFFFFFFu;
    const sid_t *vids = nullptr;
    sz = 0;
    std::vector<sid_t> out;

    if (ostat == 0 && q.opt_mode) {
        // known_to_unknown under OPTIONAL — sparql.hpp:316-375: skip
        // unmatched/BLANK rows (keep + BLANK, flag unchanged); deg-0
        // matched rows keep + BLANK with flag still true
        std::vector<uint8_t> nm;
        for (int64_t i = 0; i < R; i++) {
            sid_t cur = q.table[i * q.col_num + col];
            if (!q.matched[i] || cur == OBLANK) {
                out.insert(out.end(), q.table.begin() + i * q.col_num,
                           q.table.begin() + (i + 1) * q.col_num);
                out.push_back(OBLANK);
                nm.push_back(q.matched[i]);
                continue;
            }
            if (cur != cached) {
                cached = cur;
                if ((sid_t)p == TYPE_ID && d == DIR_IN)
                    vids = c.get(0, cur, d, &sz);
                else
                    vids = c.get(cur, (uint64_t)p, d, &sz);
            }
            if (sz > 0) {
                for (uint64_t k = 0; k < sz; k++) {
                    out.insert(out.end(), q.table.begin() + i * q.col_num,
                               q.table.begin() + (i + 1) * q.col_num);
                    out.push_back(vids[k]);
                    nm.push_back(1);
                }
            } else {
                out.insert(out.end(), q.table.begin() + i * q.col_num,
                           q.table.begin() + (i + 1) * q.col_num);
                out.push_back(OBLANK);
                nm.push_back(1);
            }
        }
        q.table.swap(out);
        q.matched.swap(nm);
        q.v2c[-(o + 1)] = q.col_num;
        q.opt_mask |= 1u << q.col_num;
        q.col_num += 1;
    } else if ((ostat == 1 || ostat == 2) && q.opt_mode) {
        // known_to_known / known_to_const under OPTIONAL: keep rows,
        // blank + clear flag on mismatch (sparql.hpp:416-549 OPTIONAL)
        for (int64_t i = 0; i < R; i++) {
            sid_t cur = q.table[i * q.col_num + col];
            bool ok = false;
            if (cur != OBLANK) {
                if (cur != cached) {
                    cached = cur;
                    vids = c.get(cur, (uint64_t)p, d, &sz);
                }
                sid_t tgt = (ostat == 2) ? (sid_t)o
                                         : q.table[i * q.col_num + q.var2col(o)];
                if (tgt != OBLANK)
                    for (uint64_t k = 0; k < sz; k++)
                        if (vids[k] == tgt) { ok = true; break; }
            }
            if (!ok) {
                if (q.matched[i]) q.blank_row(i);
                q.matched[i] = 0;
            }
        }
    } else if (ostat == 0) {
        // known_to_unknown — sparql.hpp:295-407
        for (int64_t i = 0; i < R; i++) {
            sid_t cur = q.table[i * q.col_num + col];
            if (cur != cached) {
                cached = cur;
                if ((sid_t)p == TYPE_ID && d == DIR_IN)
                    vids = c.get(0, cur, d, &sz);       // :340-341 get_index
                else
                    vids = c.get(cur, (uint64_t)p, d, &sz);
            }
            for (uint64_t k = 0; k < sz; k++) {
                out.insert(out.end(), q.table.begin() + i * q.col_num,
                           q.table.begin() + (i + 1) * q.col_num);
                out.push_back(vids[k]);
            }
        }
        q.table.swap(out);
        q.v2c[-(o + 1)] = q.col_num;
        q.col_num += 1;
    } else if (ostat == 1) {
        // known_to_known — sparql.hpp:416-476
        int col2 = q.var2col(o);
        for (int64_t i = 0; i < R; i++) {
            sid_t cur = q.table[i * q.col_num + col];
            if (cur != cached) {
                cached = cur;
                vids = c.get(cur, (uint64_t)p, d, &sz);
            }
            sid_t known = q.table[i * q.col_num + col2];
            for (uint64_t k = 0; k < sz; k++)
                if (vids[k] == known) {
                    out.insert(out.end(), q.table.begin() + i * q.col_num,
                               q.table.begin() + (i + 1) * q.col_num);
                    break;
                }
        }
        q.table.swap(out);
    } else {
        // known_to_const — sparql.hpp:484-549 (memoised exist flag)
        bool exist = false;
        for (int64_t i = 0; i < R; i++) {
            sid_t cur = q.table[i * q.col_num + col];
            if (cur != cached) {
                cached = cur;
                exist = false;
                vids = c.get(cur, (uint64_t)p, d, &sz);
                for (uint64_t k = 0; k < sz; k++)
                    if (vids[k] == (sid_t)o) { exist = true; break; }
            }
            if (exist)
                out.insert(out.end(), q.table.begin() + i * q.col_num,
                           q.table.begin() + (i + 1) * q.col_num);
        }
        q.table.swap(out);
    }
    q.step++;
}

// final ops — sparql.hpp:1424-1551 (DISTINCT = full-row sort + adjacent
// equal-on-required-vars removal, exactly the reference's algorithm)
static void final_process(query &q, const ssid_t *req, int nreq, int distinct,
                          int64_t limit, int64_t offset) {
    int C = q.col_num;
    int64_t R = q.nrows();
    if (distinct && R > 0) {
        std::vector<int64_t> idx(R);
        for (int64_t i = 0; i < R; i++) idx[i] = i;
        std::sort(idx.begin(), idx.end(), [&](int64_t a, int64_t b) {
            for (int c2 = 0; c2 < C; c2++) {
                sid_t x = q.table[a * C + c2], y = q.table[b * C + c2];
                if (x != y) return x < y;
            }
            return false;
        });
        std::vector<sid_t> kept;
        auto eq_req = [&](int64_t a, int64_t b) {
            for (int i = 0; i < nreq; i++) {
                int cc = q.var2col(req[i]);
                if (q.table[a * C + cc] != q.table[b * C + cc]) return false;
            }
            return true;
        };
        for (int64_t i = 0; i < R; i++) {
            if (i > 0 && eq_req(idx[i - 1], idx[i])) continue;
            kept.insert(kept.end(), q.table.begin() + idx[i] * C,
                        q.table.begin() + (idx[i] + 1) * C);
        }
        q.table.swap(kept);
        R = q.nrows();
    }
    if (offset > 0) {
        int64_t drop = std::min<int64_t>(offset, R);
        q.table.erase(q.table.begin(), q.table.begin() + drop * C);
        R -= drop;
    }
    if (limit >= 0 && R > limit) { q.table.resize((size_t)limit * C); R = limit; }
    // projection — sparql.hpp:1510-1536
    std::vector<sid_t> proj((size_t)R * nreq);
    for (int64_t i = 0; i < R; i++)
        for (int j = 0; j < nreq; j++)
            proj[i * nreq + j] = q.table[i * C + q.var2col(req[j])];
    q.table.swap(proj);
    q.col_num = nreq;
}

}  // namespace ok

// ---------------------------------------------------------------------
// C ABI (for ctypes in tests and bench.py cpu_baseline)
// ---------------------------------------------------------------------
extern "C" {

typedef struct {
    int32_t subject, predicate, object, direction;
} ok_pattern_t;  // matches wk_pattern_t layout

void *ok_build(const uint32_t *spo, int64_t n, int32_t sid, int32_t nsrv) {
    return ok::build(spo, n, sid, nsrv);
}
void ok_free(void *c) { delete (ok::ctx *)c; }

const uint32_t *ok_get_triples(void *cv, uint32_t vid, uint32_t pid, int32_t dir,
                               uint64_t *sz) {
    return ((ok::ctx *)cv)->get(vid, pid, dir, sz);
}
const uint32_t *ok_get_index(void *cv, uint32_t pid, int32_t dir, uint64_t *sz) {
    return ((ok::ctx *)cv)->get(0, pid, dir, sz);
}

uint64_t ok_hash_u64(uint64_t x) { return ok::hash_u64(x); }
uint64_t ok_key_pack(uint64_t vid, uint64_t pid, uint64_t dir) {
    return ok::key_pack(vid, pid, dir);
}

// Run a full plan.  mt = number of slice-threads for index-start queries
// (the reference's num_servers*mt_factor dispatch, sparql.hpp:1064-1111;
// sub-results merged in slice order).  Returns row count; caller frees
// *out with ok_free_table.
int64_t ok_run_query(void *cv, const ok_pattern_t *pats, int32_t npat,
                     int32_t nvars, const int32_t *req, int32_t nreq,
                     int32_t distinct, int64_t limit, int64_t offset,
                     int32_t mt, uint32_t **out, int32_t *out_cols) {
    ok::ctx *c = (ok::ctx *)cv;
    bool index_start = npat > 0 && pats[0].subject >= 0 && ok::is_tpid(pats[0].subject);
    int nslices = (index_start && mt > 1) ? mt : 1;

    std::vector<ok::query> qs(nslices);
    for (int t = 0; t < nslices; t++) {
        ok::query &q = qs[t];
        for (int i = 0; i < npat; i++)
            q.pats.push_back({pats[i].subject, pats[i].predicate, pats[i].object,
                              pats[i].direction});
        q.nvars = nvars;
        q.v2c.assign(nvars, -1);
        q.mt_tid = t;
        q.mt_factor = nslices;
    }
#pragma omp parallel for num_threads(nslices) schedule(static, 1)
    for (int t = 0; t < nslices; t++) {
        ok::query &q = qs[t];
        while (q.step < (int)q.pats.size()) ok::exec_pattern(*c, q);
    }
    // merge (RMap::merge/append_result, rmap.hpp:57-87, query.hpp:697-718)
    ok::query &q0 = qs[0];
    for (int t = 1; t < nslices; t++)
        q0.table.insert(q0.table.end(), qs[t].table.begin(), qs[t].table.end());

    ok::final_process(q0, req, nreq, distinct, limit, offset);
    int64_t R = q0.nrows();
    *out_cols = q0.col_num;
    *out = (uint32_t *)malloc(std::max<size_t>(q0.table.size() * 4, 4));
    memcpy(*out, q0.table.data(), q0.table.size() * 4);
    return R;
}

// Extended run: main BGP + UNION branches + OPTIONAL group
// (execute_sparql_query order, sparql.hpp:1564-1662; union merge =
// rmap.hpp:57-87 row concat; optional = matched-flag + BLANK fill).
int64_t ok_run_query_ex(void *cv, const ok_pattern_t *pats, int32_t npat,
                        int32_t nvars, const ok_pattern_t *opt, int32_t nopt,
                        const ok_pattern_t *upats, const int32_t *usizes,
                        int32_t nunion, const int32_t *req, int32_t nreq,
                        int32_t distinct, int64_t limit, int64_t offset,
                        uint32_t **out, int32_t *out_cols) {
    ok::ctx *c = (ok::ctx *)cv;
    ok::query q;
    for (int i = 0; i < npat; i++)
        q.pats.push_back({pats[i].subject, pats[i].predicate, pats[i].object,
                          pats[i].direction});
    q.nvars = nvars;
    q.v2c.assign(nvars, -1);
    while (q.step < (int)q.pats.size()) ok::exec_pattern(*c, q);

    if (nunion > 0) {
        std::vector<ok::sid_t> merged;
        std::vector<int> bv2c;
        int bc = -1;
        const ok_pattern_t *bp = upats;
        for (int b = 0; b < nunion; b++) {
            ok::query qb = q;  // branch inherits the parent result
            qb.pats.clear();
            for (int i = 0; i < usizes[b]; i++)
                qb.pats.push_back({bp[i].subject, bp[i].predicate,
                                   bp[i].object, bp[i].direction});
            bp += usizes[b];
            qb.step = 0;
            while (qb.step < (int)qb.pats.size()) ok::exec_pattern(*c, qb);
            if (bc < 0) { bc = qb.col_num; bv2c = qb.v2c; }
            merged.insert(merged.end(), qb.table.begin(), qb.table.end());
        }
        q.table.swap(merged);
        q.col_num = bc;
        q.v2c = bv2c;
    }
    if (nopt > 0) {
        q.opt_mode = 1;
        q.opt_mask = 0;
        q.matched.assign(q.nrows(), 1);
        q.pats.clear();
        for (int i = 0; i < nopt; i++)
            q.pats.push_back({opt[i].subject, opt[i].predicate, opt[i].object,
                              opt[i].direction});
        q.step = 0;
        while (q.step < (int)q.pats.size()) ok::exec_pattern(*c, q);
        q.opt_mode = 0;
    }

    ok::final_process(q, req, nreq, distinct, limit, offset);
    int64_t R = q.nrows();
    *out_cols = q.col_num;
    *out = (uint32_t *)malloc(std::max<size_t>(q.table.size() * 4, 4));
    memcpy(*out, q.table.data(), q.table.size() * 4);
    return R;
}

void ok_free_table(uint32_t *t) { free(t); }

// ---- step-level API (mirrors the engine's five-call surface so the
// distributed driver can be exercised on CPU with gloo) ----
struct ok_query_handle {
    ok::ctx *c;
    ok::query q;
};

void *ok_query_begin(void *cv, const ok_pattern_t *pats, int32_t npat, int32_t nvars) {
    ok_query_handle *h = new ok_query_handle();
    h->c = (ok::ctx *)cv;
    for (int i = 0; i < npat; i++)
        h->q.pats.push_back({pats[i].subject, pats[i].predicate, pats[i].object,
                             pats[i].direction});
    h->q.nvars = nvars;
    h->q.v2c.assign(nvars, -1);
    return h;
}

void ok_query_load(void *hv, const uint32_t *table, int64_t nrows, int32_t ncols,
                   const int32_t *v2c, int32_t step) {
    ok_query_handle *h = (ok_query_handle *)hv;
    h->q.table.assign(table, table + (size_t)nrows * ncols);
    h->q.col_num = ncols;
    h->q.step = step;
    if (v2c) h->q.v2c.assign(v2c, v2c + h->q.nvars);
}

int64_t ok_query_step(void *hv) {
    ok_query_handle *h = (ok_query_handle *)hv;
    if (h->q.step >= (int)h->q.pats.size()) return -1;
    ok::exec_pattern(*h->c, h->q);
    return h->q.nrows();
}

int32_t ok_query_cols(void *hv) { return ((ok_query_handle *)hv)->q.col_num; }
int32_t ok_query_stepno(void *hv) { return ((ok_query_handle *)hv)->q.step; }

int64_t ok_query_table(void *hv, const uint32_t **data) {
    ok_query_handle *h = (ok_query_handle *)hv;
    *data = h->q.table.data();
    return h->q.nrows();
}

void ok_query_finalize(void *hv, const int32_t *req, int32_t nreq,
                       int32_t distinct, int64_t limit, int64_t offset) {
    ok_query_handle *h = (ok_query_handle *)hv;
    ok::final_process(h->q, req, nreq, distinct, limit, offset);
}

void ok_query_free(void *hv) { delete (ok_query_handle *)hv; }

}  // extern "C"
