/*
 * ORACLE PIN — TEST INFRASTRUCTURE ONLY.
 *
 * Algorithmically independent brute-force BGP evaluator used to pin the
 * oracle: evaluates the same plans by hash-join over the
 * raw dedup'd triple list — no KV store, no predicate segments, no
 * graph exploration, no probe.  Agreement oracle==brute on seeded data
 * is the operator-level parity anchor (the reference ships no operator
 * unit tests — SURVEY.md §4/§8c).
 */
#include <cstdint>
#include <cstring>
#include <cstdlib>
#include <vector>
#include <unordered_map>
#include <unordered_set>
#include <algorithm>
#include "ok_internal.h"

namespace ok {
constexpr sid_t B_PREDICATE_ID = PREDICATE_ID;
constexpr sid_t B_TYPE_ID = TYPE_ID;
static inline bool b_is_tpid(int64_t id) { return is_tpid(id); }
}  // namespace ok

extern "C" {

typedef struct { int32_t subject, predicate, object, direction; } ok_pattern_t;

int64_t ok_brute_query(void *cv, const ok_pattern_t *pats, int32_t npat,
                       int32_t nvars, const int32_t *req, int32_t nreq,
                       uint32_t **out, int32_t *out_cols) {
    using namespace ok;
    ctx *c = (ctx *)cv;
    const auto &T = c->triples;

    std::vector<sid_t> table;
    int C = 0;
    std::vector<int> v2c(nvars, -1);
    auto col_of = [&](ssid_t v) { return v < 0 ? v2c[-(v + 1)] : -1; };

    for (int pi = 0; pi < npat; pi++) {
        const ok_pattern_t &pt = pats[pi];
        ssid_t a = pt.subject, p = pt.predicate, o = pt.object;
        int d = pt.direction;

        // directed pair list (va, vb) for this pattern
        std::vector<std::pair<sid_t, sid_t>> pairs;
        if (pi == 0 && a >= 0 && b_is_tpid(a)) {
            // index start: unary candidate set for ?o
            std::vector<sid_t> xs;
            if (p == (ssid_t)B_TYPE_ID) {
                for (const auto &t : T)
                    if (t.p == B_TYPE_ID && t.o == (sid_t)a) xs.push_back(t.s);
            } else {  // __PREDICATE__ index
                std::unordered_set<sid_t> set;
                for (const auto &t : T)
                    if (t.p == (sid_t)a) set.insert(d == 0 ? t.s : t.o);
                xs.assign(set.begin(), set.end());
            }
            table.assign(xs.begin(), xs.end());
            C = 1;
            v2c[-(o + 1)] = 0;
            continue;
        }
        // p < 0: VERSATILE (predicate variable) — match every predicate,
        // the predicate id becomes a column.  Independent restatement:
        // no predicate lists, just raw triple matching.
        if (p < 0) {
            int ostat = (o >= 0) ? 2 : (col_of(o) >= 0 ? 1 : 0);
            std::vector<sid_t> next;
            auto each = [&](auto &&f) {
                for (const auto &t : T) {
                    if (d == 1) f(t.s, t.o, t.p);
                    else f(t.o, t.s, t.p);
                }
            };
            if (a >= 0) {  // const start (first pattern)
                if (ostat == 0) {
                    each([&](sid_t va, sid_t vb, sid_t pp) {
                        if (va == (sid_t)a) {
                            next.push_back(pp);
                            next.push_back(vb);
                        }
                    });
                    table.swap(next);
                    v2c[-(p + 1)] = C;
                    v2c[-(o + 1)] = C + 1;
                    C += 2;
                } else {  // const_unknown_const: distinct linking preds
                    std::unordered_set<sid_t> preds;
                    each([&](sid_t va, sid_t vb, sid_t pp) {
                        if (va == (sid_t)a && vb == (sid_t)o) preds.insert(pp);
                    });
                    std::vector<sid_t> ps(preds.begin(), preds.end());
                    std::sort(ps.begin(), ps.end());
                    table.assign(ps.begin(), ps.end());
                    v2c[-(p + 1)] = C;
                    C += 1;
                }
                continue;
            }
            int ca = col_of(a);
            int64_t R = C ? (int64_t)table.size() / C : 0;
            if (ostat == 0) {  // known_unknown_unknown
                std::unordered_multimap<sid_t, std::pair<sid_t, sid_t>> mm;
                each([&](sid_t va, sid_t vb, sid_t pp) {
                    mm.emplace(va, std::make_pair(pp, vb));
                });
                for (int64_t i = 0; i < R; i++) {
                    auto range = mm.equal_range(table[i * C + ca]);
                    for (auto it = range.first; it != range.second; ++it) {
                        next.insert(next.end(), table.begin() + i * C,
                                    table.begin() + (i + 1) * C);
                        next.push_back(it->second.first);
                        next.push_back(it->second.second);
                    }
                }
                table.swap(next);
                v2c[-(p + 1)] = C;
                v2c[-(o + 1)] = C + 1;
                C += 2;
            } else {  // known_unknown_const: one row per distinct (va, p)
                std::unordered_set<uint64_t> vap;
                each([&](sid_t va, sid_t vb, sid_t pp) {
                    if (vb == (sid_t)o)
                        vap.insert(((uint64_t)va << 32) | pp);
                });
                for (int64_t i = 0; i < R; i++) {
                    sid_t va = table[i * C + ca];
                    // iterate candidate preds in ascending order: collect
                    std::vector<sid_t> ps;
                    for (uint64_t key : vap)
                        if ((sid_t)(key >> 32) == va)
                            ps.push_back((sid_t)(key & 0xFFFFFFFFu));
                    std::sort(ps.begin(), ps.end());
                    for (sid_t pp : ps) {
                        next.insert(next.end(), table.begin() + i * C,
                                    table.begin() + (i + 1) * C);
                        next.push_back(pp);
                    }
                }
                table.swap(next);
                v2c[-(p + 1)] = C;
                C += 1;
            }
            continue;
        }
        for (const auto &t : T) {
            if (t.p != (sid_t)p) continue;
            if (d == 1) pairs.emplace_back(t.s, t.o);
            else pairs.emplace_back(t.o, t.s);
        }

        int ostat = (o >= 0) ? 2 : (col_of(o) >= 0 ? 1 : 0);
        std::vector<sid_t> next;
        if (a >= 0) {  // const start
            if (ostat == 0) {
                for (auto &pr : pairs)
                    if (pr.first == (sid_t)a) next.push_back(pr.second);
                table.swap(next);
                v2c[-(o + 1)] = C;
                C += 1;
            } else {
                std::unordered_set<sid_t> set;
                for (auto &pr : pairs)
                    if (pr.first == (sid_t)a) set.insert(pr.second);
                int co = col_of(o);
                int64_t R = C ? (int64_t)table.size() / C : 0;
                for (int64_t i = 0; i < R; i++)
                    if (set.count(table[i * C + co]))
                        next.insert(next.end(), table.begin() + i * C,
                                    table.begin() + (i + 1) * C);
                table.swap(next);
            }
            continue;
        }
        int ca = col_of(a);
        int64_t R = C ? (int64_t)table.size() / C : 0;
        if (ostat == 0) {
            std::unordered_multimap<sid_t, sid_t> mm;
            for (auto &pr : pairs) mm.emplace(pr.first, pr.second);
            for (int64_t i = 0; i < R; i++) {
                auto range = mm.equal_range(table[i * C + ca]);
                for (auto it = range.first; it != range.second; ++it) {
                    next.insert(next.end(), table.begin() + i * C,
                                table.begin() + (i + 1) * C);
                    next.push_back(it->second);
                }
            }
            table.swap(next);
            v2c[-(o + 1)] = C;
            C += 1;
        } else {
            std::unordered_set<uint64_t> set;
            for (auto &pr : pairs)
                set.insert(((uint64_t)pr.first << 32) | pr.second);
            for (int64_t i = 0; i < R; i++) {
                sid_t va = table[i * C + ca];
                sid_t vb = (ostat == 2) ? (sid_t)o : table[i * C + col_of(o)];
                if (set.count(((uint64_t)va << 32) | vb))
                    next.insert(next.end(), table.begin() + i * C,
                                table.begin() + (i + 1) * C);
            }
            table.swap(next);
        }
    }

    // projection only (tests compare DISTINCT/LIMIT separately on the
    // oracle path; brute pins the pattern operators)
    int64_t R = C ? (int64_t)table.size() / C : 0;
    *out_cols = nreq;
    uint32_t *res = (uint32_t *)malloc(std::max<size_t>((size_t)R * nreq * 4, 4));
    for (int64_t i = 0; i < R; i++)
        for (int j = 0; j < nreq; j++)
            res[i * nreq + j] = table[i * C + col_of(req[j])];
    *out = res;
    return R;
}

}  // extern "C"
