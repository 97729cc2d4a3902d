/*
 * wk_types.h — bit-exact restatement of the reference store's key/value
 * layouts (core/store/vertex.hpp) and the TomasWang hash
 * (utils/math.hpp:58-66).  Shared by the host store builder and the HIP
 * kernels.  NOT copied: restated from the documented bit layouts.
 */
#pragma once
#include <cstdint>
#include <cstddef>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#include <hip/hip_runtime.h>
#define WK_HD __host__ __device__ inline
#else
#define WK_HD inline
#endif

namespace wk {

typedef uint32_t sid_t;   // core/type.hpp:36
typedef int32_t  ssid_t;  // core/type.hpp:37

// core/store/vertex.hpp:34-39
constexpr int NBITS_DIR = 1;
constexpr int NBITS_IDX = 17;
constexpr int NBITS_VID = 64 - NBITS_IDX - NBITS_DIR;  // 46
constexpr sid_t PREDICATE_ID = 0;
constexpr sid_t TYPE_ID = 1;
constexpr sid_t BLANK_ID = 0xFFFFFFFFu;  // core/type.hpp:38
constexpr int DIR_IN = 0, DIR_OUT = 1;   // core/store/gstore.hpp:61

// cluster-hash bucket geometry (core/store/gstore.hpp:979: ASSOCIATIVITY=8;
// slots 0..6 = data, slot 7 = chain pointer, gstore.hpp:797-828)
constexpr int ASSOC = 8;

WK_HD bool is_tpid(int64_t id) { return id > 1 && id < (1 << NBITS_IDX); }  // vertex.hpp:41
WK_HD bool is_vid(int64_t id)  { return id >= (1 << NBITS_IDX); }           // vertex.hpp:43

// ikey_t bitfields {dir:1, pid:17, vid:46} pack LSB-first on GCC/Clang:
// packed = vid<<18 | pid<<1 | dir.  This equals the hash input of
// vertex.hpp:88-96 (r = ((vid<<17)+pid)<<1 + dir), so the packed word IS
// the hash input.  Verified against the reference headers by
// oracle/ref_dump.cpp -> tests/golden/hash_golden.csv.
WK_HD uint64_t key_pack(uint64_t vid, uint64_t pid, uint64_t dir) {
    return (vid << (NBITS_IDX + NBITS_DIR)) | (pid << NBITS_DIR) | dir;
}
WK_HD uint64_t key_vid(uint64_t k) { return k >> (NBITS_IDX + NBITS_DIR); }
WK_HD uint64_t key_pid(uint64_t k) { return (k >> NBITS_DIR) & ((1ull << NBITS_IDX) - 1); }
WK_HD uint64_t key_dir(uint64_t k) { return k & 1ull; }
constexpr uint64_t KEY_EMPTY = 0;  // vertex.hpp:79 is_empty()

// iptr_t bitfields {size:28, off:34, type:2} LSB-first:
// packed = size | off<<28 | type<<62 (vertex.hpp:117-151).
WK_HD uint64_t ptr_pack(uint64_t size, uint64_t off, uint64_t type = 0) {
    return size | (off << 28) | (type << 62);
}
WK_HD uint64_t ptr_size(uint64_t p) { return p & ((1ull << 28) - 1); }
WK_HD uint64_t ptr_off(uint64_t p)  { return (p >> 28) & ((1ull << 34) - 1); }

// Rank-compressed functional-predicate map:
// one 16-B page per 64 vids of the span — a key-presence bitmap plus
// the prefix popcount (rank) — and a gap-free packed value array
// indexed by rank.  Replaces the round-1 dense vid->object array: the
// packed values of a hot LUBM segment are ~25-130 MB (Infinity-Cache
// resident) instead of a 1.3 GB sparse span, and a monotone gather
// stream touches every line fully instead of one 4-B entry per 64-B
// line.  Lookup = one 16-B page load + one 4-B value load.
struct alignas(16) fnpage_t {
    uint64_t bits;   // key-presence, bit (vid - base) & 63
    uint32_t rank;   // # keys in pages before this one
    uint32_t pad;
};

WK_HD sid_t fn_lookup(const fnpage_t *pages, const sid_t *vals,
                      uint64_t base, uint64_t n, sid_t v) {
    uint64_t idx = (uint64_t)v - base;
    if (idx >= n) return 0;
    const fnpage_t p = pages[idx >> 6];
    if (!((p.bits >> (idx & 63)) & 1)) return 0;
#if defined(__HIP_DEVICE_COMPILE__)
    uint32_t r = p.rank + (uint32_t)__popcll(p.bits & ((1ull << (idx & 63)) - 1));
#else
    uint32_t r = p.rank + (uint32_t)__builtin_popcountll(p.bits & ((1ull << (idx & 63)) - 1));
#endif
    return vals[r];
}

// 128-bit slot (vertex.hpp:154-157)
struct vertex_t { uint64_t key; uint64_t ptr; };
static_assert(sizeof(vertex_t) == 16, "slot must be 16B");

// TomasWang 64-bit mix (utils/math.hpp:58-66)
WK_HD uint64_t hash_u64(uint64_t key) {
    key = (~key) + (key << 21);
    key = key ^ (key >> 24);
    key = (key + (key << 3)) + (key << 8);
    key = key ^ (key >> 14);
    key = (key + (key << 2)) + (key << 4);
    key = key ^ (key >> 28);
    key = key + (key << 31);
    return key;
}

WK_HD uint64_t key_hash(uint64_t vid, uint64_t pid, uint64_t dir) {
    return hash_u64(key_pack(vid, pid, dir));
}

// segment descriptor: per (pid,dir) bucket range (gstore.hpp:98-120 kept
// as CSR-like contiguity; bucket allocation policy is ours)
struct seg_t {
    uint64_t bucket_start = 0;
    uint64_t num_buckets = 0;   // 0 = segment absent
};

}  // namespace wk
