/*
 * ref_dump.cpp — golden-vector dumper compiled against the REFERENCE'S
 * OWN headers (core/store/vertex.hpp + utils/math.hpp, which compile
 * standalone — SURVEY.md §8c).  Container-only: /root/reference does not
 * exist on the GPU box; the committed output tests/golden/hash_golden.csv
 * travels instead.  Build recipe: oracle/Makefile target `_ref`.
 *
 * Emits CSV rows:
 *   K,vid,pid,dir,<raw u64 of ikey_t memory>,<ikey_t::hash()>
 *   P,size,off,type,<raw u64 of iptr_t memory>
 * which pin (a) the bitfield packing of ikey_t/iptr_t and (b) the
 * TomasWang hash, against which oracle and HIP kernels are tested.
 */
#include <iostream>
#include <cassert>
#include <string>
#include <cstring>
#include <cstdio>
using namespace std;

#include "vertex.hpp"  // the reference header, via -I (see Makefile)

static uint64_t splitmix(uint64_t &s) {
    uint64_t z = (s += 0x9e3779b97f4a7c15ull);
    z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
    z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
    return z ^ (z >> 31);
}

int main() {
    // fixed edge cases + seeded randoms
    struct { uint64_t v, p, d; } cases[] = {
        {0, 0, 1}, {0, 1, 0}, {0, 1, 1}, {0, 2, 0}, {0, 131071, 1},
        {131072, 1, 1}, {131072, 2, 0}, {1, 0, 0},
        {(1ull << 46) - 1, (1ull << 17) - 1, 1},
    };
    for (auto &c : cases) {
        ikey_t k(c.v, c.p, c.d);
        uint64_t raw;
        memcpy(&raw, &k, 8);
        printf("K,%llu,%llu,%llu,%llu,%llu\n",
               (unsigned long long)c.v, (unsigned long long)c.p,
               (unsigned long long)c.d, (unsigned long long)raw,
               (unsigned long long)k.hash());
    }
    uint64_t seed = 42;
    for (int i = 0; i < 64; i++) {
        uint64_t v = splitmix(seed) & ((1ull << 46) - 1);
        uint64_t p = splitmix(seed) % ((1ull << 17) - 2) + 2;
        uint64_t d = splitmix(seed) & 1;
        ikey_t k(v, p, d);
        uint64_t raw;
        memcpy(&raw, &k, 8);
        printf("K,%llu,%llu,%llu,%llu,%llu\n",
               (unsigned long long)v, (unsigned long long)p,
               (unsigned long long)d, (unsigned long long)raw,
               (unsigned long long)k.hash());
    }
    // iptr_t packing (vertex.hpp:117-151)
    struct { uint64_t s, o, t; } pcases[] = {
        {0, 0, 0}, {1, 0, 0}, {(1ull << 28) - 1, (1ull << 34) - 1, 3},
        {12345, 678901234, 0}, {7, 1, 2},
    };
    for (auto &c : pcases) {
        iptr_t p(c.s, c.o, c.t);
        uint64_t raw;
        memcpy(&raw, &p, 8);
        printf("P,%llu,%llu,%llu,%llu\n",
               (unsigned long long)c.s, (unsigned long long)c.o,
               (unsigned long long)c.t, (unsigned long long)raw);
    }
    return 0;
}
