/*
 * wukong_abi.h — C-ABI drop-in boundary of the MI355X-native
 * graph-exploration engine.
 *
 * Each entry point cites the reference interface it replaces
 * (paths relative to the SJTU-IPADS/wukong tree).  All tables are
 * row-major uint32 (sid_t, core/type.hpp:36); variables are negative
 * ids with col = v2c_map[-(vid+1)] (core/query.hpp:352-374); constants
 * are positive sids; index ids satisfy 1 < id < 2^17, normal vertex
 * ids >= 2^17 (core/store/vertex.hpp:34-43).
 */
#ifndef WUKONG_ABI_H
#define WUKONG_ABI_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef uint32_t wk_sid_t;  /* string id            — core/type.hpp:36 */
typedef int32_t  wk_ssid_t; /* signed string id     — core/type.hpp:37 */

enum { WK_DIR_IN = 0, WK_DIR_OUT = 1 };      /* core/type.hpp dir_t    */
enum { WK_PREDICATE_ID = 0, WK_TYPE_ID = 1 };/* core/store/vertex.hpp:39 */
#define WK_BLANK_ID 0xFFFFFFFFu              /* core/type.hpp:38       */

/* One triple pattern — core/query.hpp:95-116 (Pattern{s,p,o,dir}). */
typedef struct {
    wk_ssid_t subject;
    wk_ssid_t predicate;
    wk_ssid_t object;
    int32_t   direction;  /* WK_DIR_IN / WK_DIR_OUT */
} wk_pattern_t;

/* A full query plan — the planner-ordered pattern list plus the final
 * projection (core/query.hpp required_vars/distinct/limit,
 * core/engine/sparql.hpp:1424-1551). */
typedef struct {
    const wk_pattern_t *patterns;
    int32_t   npatterns;
    int32_t   nvars;              /* variables are -1..-nvars          */
    const wk_ssid_t *required_vars;
    int32_t   nrequired;
    int32_t   distinct;           /* 0/1 */
    int64_t   limit;              /* -1 = none */
    int64_t   offset;
    int32_t   blind;              /* 1 = return row count only (Result::blind,
                                   * core/query.hpp:321 — the reference's
                                   * proxy/emulator benchmark mode) */
    /* OPTIONAL pattern group (BGP-only, matching the reference's own
     * limit — query.hpp:722-733; unmatched rows keep BLANK_ID in the
     * optional-bound columns, sparql.hpp:100-170,316-375).  NULL/0 when
     * absent. */
    const wk_pattern_t *opt_patterns;
    int32_t   nopt;
    /* UNION branches (query.hpp:708-718; sparql.hpp:1593-1614): the
     * branch pattern groups concatenated into one array, with
     * union_sizes[i] patterns per branch.  Each branch continues from
     * the main-group result table; the final result is the row concat
     * of the branch results (rmap.hpp:57-87).  All branches must bind
     * the same variable->column layout.  NULL/0 when absent. */
    const wk_pattern_t *union_pats;
    const int32_t *union_sizes;
    int32_t   nunion;
} wk_plan_t;

/* A materialised binding table (SPARQLQuery::Result subset —
 * core/query.hpp:312-334).  Owned by the library; free with
 * wk_result_free. */
typedef struct {
    int32_t   col_num;
    int64_t   row_num;
    wk_sid_t *table;       /* row-major, row_num*col_num entries */
    int32_t   status_code; /* 0 = SUCCESS (utils/errors.hpp)     */
} wk_result_t;

/* ---------- synthetic data ------------------------------------- */
/* Seeded LUBM-shaped ID-triple generator (replaces datagen/ +
 * the absent Java LUBM generator; ID scheme per
 * datagen/generate_data.cpp:122-123: index ids from 2, normal ids
 * from 2^17, fixed schema enumeration order).  Returns the triples
 * of partition `sid` of `nsrv` (pso-side: s%nsrv==sid OR pos-side:
 * o%nsrv==sid — core/loader/base_loader.hpp:344-352 keeps both).
 * Caller frees with wk_free_triples. */
int64_t wk_lubm_gen(int32_t nuniv, uint64_t seed, int32_t sid, int32_t nsrv,
                    wk_sid_t **out_spo /* 3*n entries */);
/* WatDiv-shaped generator (configs[3]); ~55 triples per product. */
int64_t wk_watdiv_gen(int64_t nproducts, uint64_t seed, int32_t sid,
                      int32_t nsrv, wk_sid_t **out_spo);
void    wk_free_triples(wk_sid_t *spo);

/* ---------- store ---------------------------------------------- */
/* Host-side store build from ID-triples (StaticGStore::init,
 * core/store/static_gstore.hpp:383-454; loader sort/dedup,
 * core/loader/base_loader.hpp:302-373). */
typedef struct wk_store wk_store_t;
wk_store_t *wk_store_build(const wk_sid_t *spo, int64_t ntriples,
                           int32_t sid, int32_t nsrv);
void        wk_store_free(wk_store_t *);

/* DGraph::get_triples / get_index (core/dgraph.hpp:106-112;
 * GStore::get_edges core/store/gstore.hpp:1043-1054).  Pointer into
 * the store's host edge array; not owned by the caller; empty =
 * (NULL, *sz==0).  Lock-free after build. */
const wk_sid_t *wk_store_get_triples(const wk_store_t *, wk_sid_t vid,
                                     wk_sid_t pid, int32_t dir, uint64_t *sz);
const wk_sid_t *wk_store_get_index(const wk_store_t *, wk_sid_t pid,
                                   int32_t dir, uint64_t *sz);

/* Store introspection (for tests / upload verification — the `gsck`
 * idea, core/store/gchecker.hpp:364-392). */
uint64_t wk_store_num_slots(const wk_store_t *);
uint64_t wk_store_num_edges(const wk_store_t *);
/* Per-(pid,dir) segment statistics — the planner's cost-model inputs
 * (the reference's type-centric stats, core/optimizer/stats.hpp):
 * distinct keys and total edges of the segment. */
int32_t wk_store_seg_stats(const wk_store_t *, uint32_t pid, int32_t dir,
                           uint64_t *keys, uint64_t *edges);
uint64_t wk_store_checksum(const wk_store_t *); /* FNV over vertices+edges */
/* Memory-usage report (GStore::print_mem_usage, gstore.hpp:1062-1103):
 * cluster-hash slots, edge arrays, and the side indexes (type_of, vp
 * CSR, functional maps, type bitmaps). */
int32_t wk_store_mem_usage(const wk_store_t *, uint64_t *slots_bytes,
                           uint64_t *edges_bytes,
                           uint64_t *side_index_bytes);
/* Full-store integrity scan (the `gsck` command, core/store/
 * gchecker.hpp:364-392): 0 = consistent, else #violations. */
uint64_t wk_store_check(const wk_store_t *);

/* ---------- GPU engine ------------------------------------------ */
/* Mirrors the five-call GPU surface: load_result_buf
 * (core/gpu/gpu_engine.hpp:253-261), execute_one_pattern (:263-336),
 * generate_sub_query (:338-391), result fetch
 * (core/gpu/gpu_engine_cuda.hpp:189-195), plus engine create =
 * GPUMem+GPUCache init (core/wukong.cpp:235-241) with the cache
 * replaced by a fully HBM-resident store. */
typedef struct wk_engine wk_engine_t;
typedef struct wk_gpu_store wk_gpu_store_t;

/* Device-resident store image shared by any number of engines on one
 * GPU (one HBM upload; engines carry only scratch) — replaces the
 * reference's per-agent GPUCache (core/gpu/gpu_cache.hpp). */
wk_gpu_store_t *wk_gpu_store_create(const wk_store_t *, int32_t device);
void            wk_gpu_store_destroy(wk_gpu_store_t *);
wk_engine_t    *wk_engine_create_on(wk_gpu_store_t *);

wk_engine_t *wk_engine_create(const wk_store_t *, int32_t device);
void         wk_engine_destroy(wk_engine_t *);

/* Enqueue a whole plan as one asynchronous launch chain (no sync) —
 * pipelined multi-engine execution = the reference proxy's in-flight
 * window (core/proxy.hpp:477-525).  Harvest with wk_engine_fetch_*. */
int32_t wk_engine_submit(wk_engine_t *, const wk_plan_t *);

/* Batched light-query window: one asynchronous launch executes n
 * queries of the dominant light-template shape (const_to_unknown +
 * rdf:type constant filter — the emulator's A1/A2/A3/A5,
 * core/proxy.hpp:391-545), one wavefront workgroup per query.  SoA
 * inputs: subject constants, predicates, directions, type filter
 * constants.  Removes the kernel-dispatch-rate wall at 1024 in-flight
 * light queries.  WK_ERR_PLAN = store has no complete single-type
 * index; fall back to per-query wk_engine_submit. */
int32_t wk_engine_submit_light_batch(wk_engine_t *, const int64_t *subj,
                                     const int32_t *pred,
                                     const int32_t *dir,
                                     const uint32_t *cval, int32_t n);
/* Batched whole-plan light queries (the LDS plan interpreter): n
 * same-template queries, ONE launch; each wavefront workgroup
 * interprets the compiled plan (const-start, then typeof/expand/
 * filter patterns) with its binding table staged in LDS, covering the
 * emulator's multi-pattern templates A4/A6.  consts[i] replaces
 * patterns[0].subject.  Blind replies only; a query whose table
 * outgrows LDS reports count UINT64_MAX and must be re-run on the
 * per-pattern path.  WK_ERR_PLAN = template shape not interpretable. */
int32_t wk_engine_submit_plan_batch(wk_engine_t *, const wk_plan_t *tmpl,
                                    const int64_t *consts, int32_t n);
/* Blocks until the window completes; fills the n per-query row counts
 * (blind replies, Result::blind — proxy.hpp:491). */
int32_t wk_engine_light_batch_wait(wk_engine_t *, uint64_t *counts,
                                   int32_t n);

/* hipGraph replay of a whole fixed plan: build captures the submit
 * launch chain once (after a warm pass settles capacities); run
 * replays it in ONE hipGraphLaunch and returns the blind row count.
 * WK_ERR_CAP from run = capacity overflow (data changed since build):
 * fall back to wk_engine_submit. */
int32_t wk_engine_graph_build(wk_engine_t *, const wk_plan_t *,
                              int32_t *graph_id);
int32_t wk_engine_graph_run(wk_engine_t *, int32_t graph_id,
                            int64_t *nrows);
/* Asynchronous replay (no sync): back-to-back graphs on one stream
 * serialize safely, so a whole suite pass costs ONE wk_engine_sync —
 * the reference proxy's in-flight window applied to replays. */
/* Whole-suite capture: N plans into ONE instantiated graph (pays the
 * graph-replay floor once per pass).  Same gid space; replay with
 * wk_engine_graph_launch + wk_engine_sync. */
int32_t wk_engine_graph_build_suite(wk_engine_t *, const wk_plan_t *plans,
                                    int32_t nplans, int32_t *gid);
int32_t wk_engine_graph_launch(wk_engine_t *, int32_t graph_id);
int32_t wk_engine_sync(wk_engine_t *);

/* Whole-query execution on one GPU (Engine::execute_sparql_query +
 * SPARQLEngine::execute_patterns, core/engine/sparql.hpp:1113-1154,
 * 1564-1672, single-server path).  Returns 0 on success. */
int32_t wk_engine_run_query(wk_engine_t *, const wk_plan_t *, wk_result_t *out);

/* Step-level API (multi-GPU driver / tests). The engine holds ONE
 * current query state with a dual device rbuf (gpu_mem.hpp:116-124). */
int32_t wk_engine_begin_query(wk_engine_t *, const wk_plan_t *);
/* Upload/replace the current binding table (H2D load_result_buf). */
int32_t wk_engine_load_rbuf(wk_engine_t *, const wk_sid_t *table,
                            int64_t nrows, int32_t ncols,
                            const int32_t *v2c_map, int32_t pattern_step);
/* Device-pointer variant: table already in HBM (e.g. a torch tensor). */
int32_t wk_engine_load_rbuf_device(wk_engine_t *, const wk_sid_t *dev_table,
                                   int64_t nrows, int32_t ncols,
                                   const int32_t *v2c_map, int32_t pattern_step);
/* Run pattern [pattern_step]; advances the step.  Out: new row count.
 * nrows_out = NULL launches asynchronously (no host sync; the overflow
 * re-run then resolves at the next synchronizing call — only safe for
 * steps that cannot overflow, i.e. filters). */
int32_t wk_engine_execute_one_pattern(wk_engine_t *, int64_t *nrows_out);
/* Sync and return the current row count (pairs with async steps). */
int32_t wk_engine_row_count(wk_engine_t *, int64_t *nrows_out);
int32_t wk_engine_pattern_step(const wk_engine_t *);
int32_t wk_engine_col_num(const wk_engine_t *);
/* Execute the CURRENT pattern (a mid-plan const-/index-start membership
 * filter, sparql.hpp:80-186) against a caller-supplied SORTED edge list
 * instead of the local store.  Distributed driver use: the constant's
 * edge list lives only on rank `const % nsrv`, so the driver broadcasts
 * the owner's list first (the reference reads it in place over
 * one-sided RDMA, gstore.hpp:260-338).  Advances the step. */
int32_t wk_engine_execute_filter_list(wk_engine_t *, const wk_sid_t *sorted_list,
                                      uint64_t n, int64_t *nrows_out);

/* ---- xGMI peer mappings (small-table remote reads) ------------------
 * The reference reads sub-threshold tables' remote edge lists in place
 * over one-sided RDMA instead of fork-joining (need_fork_join,
 * core/engine/sparql.hpp:802-814; GStore::get_edges remote leg,
 * core/store/gstore.hpp:260-338).  The MI355X equivalent: each rank
 * exports its HBM store image as HIP-IPC handles; every other rank maps
 * them once over xGMI and k_peer_step probes the owner's store
 * directly.  The driver exchanges blobs + segment tables host-side
 * (torch.distributed), then calls import. */
#define WK_IPC_HANDLE_BYTES 64
typedef struct {
    int32_t device, sid, nsrv;
    int32_t has_type_of;
    uint64_t type_base, type_n;
    uint8_t verts_h[WK_IPC_HANDLE_BYTES];
    uint8_t edges_h[WK_IPC_HANDLE_BYTES];
    uint8_t type_of_h[WK_IPC_HANDLE_BYTES];
    int64_t nseg; /* seg-table entries ({bucket_start, num_buckets}) */
} wk_peer_blob_t;

int32_t wk_gpu_store_export(wk_gpu_store_t *, wk_peer_blob_t *out);
/* Flattened segment table: nseg x {bucket_start, num_buckets} u64 pairs.
 * Returns entry count; out=NULL queries the size. */
int64_t wk_store_seg_table(const wk_store_t *, uint64_t *out,
                           int64_t cap_entries);
int32_t wk_gpu_store_import_peers(wk_gpu_store_t *, const wk_peer_blob_t *blobs,
                                  const uint64_t *segtabs, int64_t nseg,
                                  int32_t nsrv);
/* Run the CURRENT pattern via the peer-probe path (tables below the
 * fork-join threshold).  Rejects const-/index-start and per-row
 * type-index shapes (WK_ERR_PLAN): callers fall back to the exchange. */
int32_t wk_engine_execute_one_pattern_remote(wk_engine_t *, int64_t *nrows_out);
/* Split the current table by hash of the next pattern's start var
 * (sparql.hpp:746-799; gpu_hash.cu:600-760): fills per-destination row
 * counts and packs rows into dev_out (device buffer, row-major,
 * chunks contiguous per destination, capacity cap_rows rows). */
int32_t wk_engine_generate_sub_query(wk_engine_t *, int32_t ndst,
                                     wk_sid_t *dev_out, int64_t cap_rows,
                                     int64_t *rows_per_dst);
/* Finish: run remaining host-side final ops and download. */
int32_t wk_engine_fetch_result(wk_engine_t *, const wk_plan_t *, wk_result_t *out);
/* Download the CURRENT binding table without final ops (tests/driver). */
int32_t wk_engine_fetch_raw(wk_engine_t *, wk_result_t *out);

void wk_result_free(wk_result_t *);

/* Per-kernel timing of the last run_query (HIP events, engine stream):
 * totals in microseconds + bytes of algorithmic traffic,
 * slots: 0=probe,1=scan,2=expand,3=filter,4=i2u/c2u,5=split,6=other. */
int32_t wk_engine_kernel_stats(wk_engine_t *, double *usec7, double *bytes7,
                               int64_t *launches7);

/* Raw device allocation (tests / exchange buffers). */
void   *wk_dev_alloc(uint64_t bytes);
void    wk_dev_free(void *);
int32_t wk_dev_download(const void *dev, void *host, uint64_t bytes);

/* Library/build info: returns gfx arch string the .so was built for. */
const char *wk_build_arch(void);
int32_t     wk_device_count(void);

/* Layout pinning (tests only): the reference's TomasWang hash
 * (utils/math.hpp:58-66) and ikey_t/iptr_t bit packings
 * (core/store/vertex.hpp:50-151), checked against golden vectors dumped
 * from the reference's own headers (tests/golden/hash_golden.csv). */
uint64_t wk_hash_u64(uint64_t x);
uint64_t wk_key_pack(uint64_t vid, uint64_t pid, uint64_t dir);
uint64_t wk_ptr_pack(uint64_t size, uint64_t off, uint64_t type);

#ifdef __cplusplus
}
#endif
#endif /* WUKONG_ABI_H */
