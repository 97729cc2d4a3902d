/*
 * watdiv_gen.cpp — seeded WatDiv-shaped synthetic ID-triple generator
 * (BASELINE.json configs[3]: "WatDiv-1B ... star/linear/snowflake
 * templates (stress hash-join/prune)").
 *
 * Mirrors the WatDiv core schema's join structure — products with
 * genre hubs, offers/reviews fanning into products, retailer hubs,
 * user friendship Zipf tails — not its string vocabulary.  Triples per
 * product ~= 55, so WatDiv-1B ~= 18M products.  All ids are closed-form
 * (deterministic, partition-independent), same ID scheme as LUBM
 * (index ids < 2^17, entities >= 2^17 — datagen/generate_data.cpp:122).
 */
#include "wk_types.h"
#include <vector>
#include <cstdlib>
#include <cstring>
#include <omp.h>

using namespace wk;

namespace watdiv {

enum : sid_t {
    // predicate ids (shared id space with types, from 2)
    P_HASGENRE = 2,       // product -> genre (250 hub objects)
    P_OFFER_PRODUCT = 3,  // offer -> product
    P_RETAILER = 4,       // offer -> retailer (1000 hub objects)
    P_REVIEW_PRODUCT = 5, // review -> product
    P_REVIEWER = 6,       // review -> user
    P_PURCHASED = 7,      // user -> product
    P_FRIEND = 8,         // user -> user (Zipf out-degree)
    T_PRODUCT = 9,
    T_OFFER = 10,
    T_REVIEW = 11,
    T_USER = 12,
    T_GENRE = 13,
    T_RETAILER = 14,
};

constexpr uint64_t BASE = 1ull << NBITS_IDX;
constexpr uint32_t NGENRE = 250, NRETAILER = 1000;

struct rng_t {
    uint64_t s;
    uint64_t next() {
        uint64_t z = (s += 0x9e3779b97f4a7c15ull);
        z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
        z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
        return z ^ (z >> 31);
    }
    uint32_t range(uint32_t lo, uint32_t hi) { return lo + (uint32_t)(next() % (hi - lo)); }
};

struct layout {
    uint64_t nprod, noffer, nreview, nuser;
    uint64_t genre0 = BASE, retailer0 = BASE + NGENRE;
    uint64_t prod0, offer0, review0, user0, end;
    explicit layout(uint64_t np) {
        nprod = np;
        noffer = np * 10;
        nreview = np * 3;
        nuser = np / 2 + 1;
        prod0 = retailer0 + NRETAILER;
        offer0 = prod0 + nprod;
        review0 = offer0 + noffer;
        user0 = review0 + nreview;
        end = user0 + nuser;
    }
};

struct emitter {
    std::vector<sid_t> &out;
    int sid, nsrv;
    void emit(uint64_t s, sid_t p, uint64_t o) {
        if ((int)(s % (uint64_t)nsrv) == sid || (int)(o % (uint64_t)nsrv) == sid) {
            out.push_back((sid_t)s); out.push_back(p); out.push_back((sid_t)o);
        }
    }
};

}  // namespace watdiv

extern "C" int64_t wk_watdiv_gen(int64_t nproducts, uint64_t seed, int32_t sid,
                                 int32_t nsrv, sid_t **out_spo) {
    using namespace watdiv;
    if (nproducts <= 0 || nsrv <= 0 || sid < 0 || sid >= nsrv || !out_spo)
        return -1;
    layout L((uint64_t)nproducts);
    if (L.end >= 0xFFFFFFFFull) return -1;  // u32 sid space

    const int CH = 64;  // parallel chunks
    std::vector<std::vector<sid_t>> parts(CH);
#pragma omp parallel for schedule(dynamic)
    for (int c = 0; c < CH; c++) {
        emitter em{parts[c], sid, nsrv};
        rng_t rng{hash_u64(seed * 1315423911ull + c + 1)};
        auto span = [&](uint64_t n) {
            uint64_t lo = n * c / CH, hi = n * (c + 1) / CH;
            return std::make_pair(lo, hi);
        };
        if (c == 0) {  // hub entities
            for (uint32_t g = 0; g < NGENRE; g++)
                em.emit(L.genre0 + g, TYPE_ID, T_GENRE);
            for (uint32_t r = 0; r < NRETAILER; r++)
                em.emit(L.retailer0 + r, TYPE_ID, T_RETAILER);
        }
        auto [p0, p1] = span(L.nprod);
        for (uint64_t p = p0; p < p1; p++) {
            uint64_t id = L.prod0 + p;
            em.emit(id, TYPE_ID, T_PRODUCT);
            int ng = 1 + (int)(rng.next() % 3);
            for (int i = 0; i < ng; i++)
                em.emit(id, P_HASGENRE, L.genre0 + rng.range(0, NGENRE));
        }
        auto [o0, o1] = span(L.noffer);
        for (uint64_t o = o0; o < o1; o++) {
            uint64_t id = L.offer0 + o;
            em.emit(id, TYPE_ID, T_OFFER);
            em.emit(id, P_OFFER_PRODUCT, L.prod0 + rng.next() % L.nprod);
            em.emit(id, P_RETAILER, L.retailer0 + rng.range(0, NRETAILER));
        }
        auto [r0, r1] = span(L.nreview);
        for (uint64_t r = r0; r < r1; r++) {
            uint64_t id = L.review0 + r;
            em.emit(id, TYPE_ID, T_REVIEW);
            em.emit(id, P_REVIEW_PRODUCT, L.prod0 + rng.next() % L.nprod);
            em.emit(id, P_REVIEWER, L.user0 + rng.next() % L.nuser);
        }
        auto [u0, u1] = span(L.nuser);
        for (uint64_t u = u0; u < u1; u++) {
            uint64_t id = L.user0 + u;
            em.emit(id, TYPE_ID, T_USER);
            int np = (int)(rng.next() % 21);
            for (int i = 0; i < np; i++)
                em.emit(id, P_PURCHASED, L.prod0 + rng.next() % L.nprod);
            // Zipf-ish friend out-degree: 1/(k) tail capped at 50
            int nf = (int)(50.0 / (1 + rng.next() % 50));
            for (int i = 0; i < nf; i++)
                em.emit(id, P_FRIEND, L.user0 + rng.next() % L.nuser);
        }
    }
    int64_t total = 0;
    std::vector<int64_t> offs(CH + 1, 0);
    for (int c = 0; c < CH; c++) { offs[c] = total; total += (int64_t)parts[c].size(); }
    sid_t *buf = (sid_t *)malloc((size_t)total * sizeof(sid_t));
    if (!buf) return -1;
#pragma omp parallel for schedule(dynamic)
    for (int c = 0; c < CH; c++)
        memcpy(buf + offs[c], parts[c].data(), parts[c].size() * sizeof(sid_t));
    *out_spo = buf;
    return total / 3;
}
