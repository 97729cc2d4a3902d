/*
 * ORACLE internals — TEST INFRASTRUCTURE ONLY (see oracle.cpp header).
 * Shared context type between the oracle engine (oracle.cpp) and the
 * independent brute-force pin (brute.cpp).
 */
#pragma once
#include <cstdint>
#include <vector>
#include <unordered_map>

namespace ok {

typedef uint32_t sid_t;
typedef int32_t ssid_t;

constexpr int NBITS_IDX = 17;
constexpr sid_t PREDICATE_ID = 0, TYPE_ID = 1;
constexpr int DIR_IN = 0, DIR_OUT = 1;

static inline bool is_tpid(int64_t id) { return id > 1 && id < (1 << NBITS_IDX); }

// ikey_t packing — core/store/vertex.hpp:50-97 ({dir:1,pid:17,vid:46})
static inline uint64_t key_pack(uint64_t vid, uint64_t pid, uint64_t dir) {
    return (vid << 18) | (pid << 1) | dir;
}

struct triple { sid_t s, p, o; };

struct ctx {
    std::vector<triple> triples;  // dedup'd pso-side input (brute.cpp)
    std::vector<sid_t> edges;     // all edge lists
    // kv sharded by a key mix so the build can insert shards in
    // parallel (bench.py's cpu_baseline builds this at LUBM-2560);
    // lookup semantics unchanged
    static constexpr int NSHARD = 64;
    std::unordered_map<uint64_t, std::pair<uint64_t, uint64_t>> kv[NSHARD];
    int sid = 0, nsrv = 1;

    static inline int shard_of(uint64_t key) {
        return (int)((key * 0x9e3779b97f4a7c15ull) >> 58);
    }

    const sid_t *get(uint64_t vid, uint64_t pid, int dir, uint64_t *sz) const {
        uint64_t k = key_pack(vid, pid, (uint64_t)dir);
        auto &m = kv[shard_of(k)];
        auto it = m.find(k);
        if (it == m.end()) { *sz = 0; return nullptr; }
        *sz = it->second.second;
        return edges.data() + it->second.first;
    }
};

}  // namespace ok
