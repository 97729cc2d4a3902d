"""wukong_amd — MI355X-native SPARQL graph-exploration engine.

Python is plumbing only: ctypes over the C-ABI boundary declared in
include/wukong_abi.h.  The compute path is the HIP
library; there is NO Python/CPU fallback — a missing .so raises.
"""
import ctypes
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_HERE, "libwukong_hip.so")

if not os.path.exists(_SO):
    raise ImportError(
        f"wukong_amd: native engine {_SO} not built. "
        "Run `python -m wukong_amd.build` (or __graft_entry__.build())."
    )
_lib = ctypes.CDLL(_SO)


class WkPattern(ctypes.Structure):
    _fields_ = [("subject", ctypes.c_int32), ("predicate", ctypes.c_int32),
                ("object", ctypes.c_int32), ("direction", ctypes.c_int32)]


class WkPlan(ctypes.Structure):
    _fields_ = [("patterns", ctypes.POINTER(WkPattern)),
                ("npatterns", ctypes.c_int32),
                ("nvars", ctypes.c_int32),
                ("required_vars", ctypes.POINTER(ctypes.c_int32)),
                ("nrequired", ctypes.c_int32),
                ("distinct", ctypes.c_int32),
                ("limit", ctypes.c_int64),
                ("offset", ctypes.c_int64),
                ("blind", ctypes.c_int32),
                # OPTIONAL group + UNION branches (wukong_abi.h)
                ("opt_patterns", ctypes.POINTER(WkPattern)),
                ("nopt", ctypes.c_int32),
                ("union_pats", ctypes.POINTER(WkPattern)),
                ("union_sizes", ctypes.POINTER(ctypes.c_int32)),
                ("nunion", ctypes.c_int32)]


class WkResult(ctypes.Structure):
    _fields_ = [("col_num", ctypes.c_int32), ("row_num", ctypes.c_int64),
                ("table", ctypes.POINTER(ctypes.c_uint32)),
                ("status_code", ctypes.c_int32)]


def _sig(name, res, args):
    f = getattr(_lib, name)
    f.restype = res
    f.argtypes = args
    return f

c_u32p = ctypes.POINTER(ctypes.c_uint32)
c_i32 = ctypes.c_int32
c_i64 = ctypes.c_int64
c_u64 = ctypes.c_uint64
c_vp = ctypes.c_void_p

_lubm_gen = _sig("wk_lubm_gen", c_i64, [c_i32, c_u64, c_i32, c_i32, ctypes.POINTER(c_u32p)])
_watdiv_gen = _sig("wk_watdiv_gen", c_i64, [c_i64, c_u64, c_i32, c_i32, ctypes.POINTER(c_u32p)])
_free_triples = _sig("wk_free_triples", None, [c_u32p])
_store_build = _sig("wk_store_build", c_vp, [c_u32p, c_i64, c_i32, c_i32])
_store_free = _sig("wk_store_free", None, [c_vp])
_get_triples = _sig("wk_store_get_triples", c_u32p, [c_vp, ctypes.c_uint32, ctypes.c_uint32, c_i32, ctypes.POINTER(c_u64)])
_get_index = _sig("wk_store_get_index", c_u32p, [c_vp, ctypes.c_uint32, c_i32, ctypes.POINTER(c_u64)])
_num_slots = _sig("wk_store_num_slots", c_u64, [c_vp])
_num_edges = _sig("wk_store_num_edges", c_u64, [c_vp])
_checksum = _sig("wk_store_checksum", c_u64, [c_vp])
_seg_stats = _sig("wk_store_seg_stats", c_i32,
                  [c_vp, ctypes.c_uint32, c_i32, ctypes.POINTER(c_u64),
                   ctypes.POINTER(c_u64)])
_mem_usage = _sig("wk_store_mem_usage", c_i32,
                  [c_vp, ctypes.POINTER(c_u64), ctypes.POINTER(c_u64),
                   ctypes.POINTER(c_u64)])
_store_check = _sig("wk_store_check", c_u64, [c_vp])
_eng_create = _sig("wk_engine_create", c_vp, [c_vp, c_i32])
_gstore_create = _sig("wk_gpu_store_create", c_vp, [c_vp, c_i32])
_gstore_destroy = _sig("wk_gpu_store_destroy", None, [c_vp])
_eng_create_on = _sig("wk_engine_create_on", c_vp, [c_vp])
_eng_submit = _sig("wk_engine_submit", c_i32, [c_vp, ctypes.POINTER(WkPlan)])
_eng_destroy = _sig("wk_engine_destroy", None, [c_vp])
_eng_submit_lb = _sig("wk_engine_submit_light_batch", c_i32,
                      [c_vp, ctypes.POINTER(c_i64), ctypes.POINTER(c_i32),
                       ctypes.POINTER(c_i32), ctypes.POINTER(ctypes.c_uint32),
                       c_i32])
_eng_lb_wait = _sig("wk_engine_light_batch_wait", c_i32,
                    [c_vp, ctypes.POINTER(c_u64), c_i32])
_eng_submit_pb = _sig("wk_engine_submit_plan_batch", c_i32,
                      [c_vp, ctypes.POINTER(WkPlan), ctypes.POINTER(c_i64),
                       c_i32])
_eng_graph_build = _sig("wk_engine_graph_build", c_i32,
                        [c_vp, ctypes.POINTER(WkPlan), ctypes.POINTER(c_i32)])
_eng_graph_run = _sig("wk_engine_graph_run", c_i32,
                      [c_vp, c_i32, ctypes.POINTER(c_i64)])
_eng_graph_launch = _sig("wk_engine_graph_launch", c_i32, [c_vp, c_i32])
_eng_graph_suite = _sig("wk_engine_graph_build_suite", c_i32,
                        [c_vp, ctypes.POINTER(WkPlan), c_i32,
                         ctypes.POINTER(c_i32)])
_eng_sync = _sig("wk_engine_sync", c_i32, [c_vp])

#: plan-batch count sentinel: table outgrew LDS, re-run per-pattern
LP_OVERFLOW = 0xFFFFFFFFFFFFFFFF
_eng_run = _sig("wk_engine_run_query", c_i32, [c_vp, ctypes.POINTER(WkPlan), ctypes.POINTER(WkResult)])
_eng_begin = _sig("wk_engine_begin_query", c_i32, [c_vp, ctypes.POINTER(WkPlan)])
_eng_load = _sig("wk_engine_load_rbuf", c_i32, [c_vp, c_u32p, c_i64, c_i32, ctypes.POINTER(c_i32), c_i32])
_eng_load_dev = _sig("wk_engine_load_rbuf_device", c_i32, [c_vp, c_vp, c_i64, c_i32, ctypes.POINTER(c_i32), c_i32])
_eng_step = _sig("wk_engine_execute_one_pattern", c_i32, [c_vp, ctypes.POINTER(c_i64)])
_eng_filter_list = _sig("wk_engine_execute_filter_list", c_i32,
                        [c_vp, c_u32p, c_u64, ctypes.POINTER(c_i64)])
_eng_step_remote = _sig("wk_engine_execute_one_pattern_remote", c_i32,
                        [c_vp, ctypes.POINTER(c_i64)])
_eng_row_count = _sig("wk_engine_row_count", c_i32,
                      [c_vp, ctypes.POINTER(c_i64)])


class WkPeerBlob(ctypes.Structure):
    _fields_ = [("device", c_i32), ("sid", c_i32), ("nsrv", c_i32),
                ("has_type_of", c_i32),
                ("type_base", c_u64), ("type_n", c_u64),
                ("verts_h", ctypes.c_uint8 * 64),
                ("edges_h", ctypes.c_uint8 * 64),
                ("type_of_h", ctypes.c_uint8 * 64),
                ("nseg", c_i64)]


_gstore_export = _sig("wk_gpu_store_export", c_i32,
                      [c_vp, ctypes.POINTER(WkPeerBlob)])
_seg_table = _sig("wk_store_seg_table", c_i64,
                  [c_vp, ctypes.POINTER(c_u64), c_i64])
_gstore_import = _sig("wk_gpu_store_import_peers", c_i32,
                      [c_vp, ctypes.POINTER(WkPeerBlob),
                       ctypes.POINTER(c_u64), c_i64, c_i32])
_eng_pattern_step = _sig("wk_engine_pattern_step", c_i32, [c_vp])
_eng_col_num = _sig("wk_engine_col_num", c_i32, [c_vp])
_eng_subq = _sig("wk_engine_generate_sub_query", c_i32, [c_vp, c_i32, c_vp, c_i64, ctypes.POINTER(c_i64)])
_eng_fetch = _sig("wk_engine_fetch_result", c_i32, [c_vp, ctypes.POINTER(WkPlan), ctypes.POINTER(WkResult)])
_eng_fetch_raw = _sig("wk_engine_fetch_raw", c_i32, [c_vp, ctypes.POINTER(WkResult)])
_res_free = _sig("wk_result_free", None, [ctypes.POINTER(WkResult)])
_kstats = _sig("wk_engine_kernel_stats", c_i32, [c_vp, ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_double), ctypes.POINTER(c_i64)])
_arch = _sig("wk_build_arch", ctypes.c_char_p, [])
_devcount = _sig("wk_device_count", c_i32, [])
_dev_alloc = _sig("wk_dev_alloc", c_vp, [c_u64])
_dev_free = _sig("wk_dev_free", None, [c_vp])
_dev_download = _sig("wk_dev_download", c_i32, [c_vp, c_vp, c_u64])


def dev_alloc(nbytes):
    p = _dev_alloc(nbytes)
    if not p:
        raise RuntimeError("wk_dev_alloc failed")
    return p


def dev_free(p):
    _dev_free(p)


def dev_download_u32(ptr, count):
    out = np.empty(count, dtype=np.uint32)
    rc = _dev_download(ptr, out.ctypes.data_as(c_vp), count * 4)
    if rc != 0:
        raise RuntimeError("wk_dev_download failed")
    return out


hash_u64 = _sig("wk_hash_u64", c_u64, [c_u64])
key_pack = _sig("wk_key_pack", c_u64, [c_u64, c_u64, c_u64])
ptr_pack = _sig("wk_ptr_pack", c_u64, [c_u64, c_u64, c_u64])

DIR_IN, DIR_OUT = 0, 1
KERNEL_CATS = ["probe", "scan", "expand", "filter", "copy", "split", "other"]


def build_arch():
    return _arch().decode()


def device_count():
    return _devcount()


def lubm_gen(nuniv, seed=42, sid=0, nsrv=1):
    """Seeded LUBM-shaped synthetic ID-triples, partition (sid of nsrv)."""
    out = c_u32p()
    n = _lubm_gen(nuniv, seed, sid, nsrv, ctypes.byref(out))
    if n < 0:
        raise RuntimeError("wk_lubm_gen failed")
    arr = np.ctypeslib.as_array(out, shape=(n, 3)).copy()
    _free_triples(out)
    return arr


def watdiv_gen(nproducts, seed=42, sid=0, nsrv=1):
    """Seeded WatDiv-shaped synthetic ID-triples (~55 triples/product)."""
    out = c_u32p()
    n = _watdiv_gen(nproducts, seed, sid, nsrv, ctypes.byref(out))
    if n < 0:
        raise RuntimeError("wk_watdiv_gen failed")
    arr = np.ctypeslib.as_array(out, shape=(n, 3)).copy()
    _free_triples(out)
    return arr


class Plan:
    """A planner-ordered pattern list (vars negative: -1..-nvars)."""

    def __init__(self, patterns, nvars, required_vars, distinct=False,
                 limit=-1, offset=0, blind=False, optional=None, unions=None):
        self.patterns = list(patterns)
        self.nvars = nvars
        self.required_vars = list(required_vars)
        self.distinct = distinct
        self.limit = limit
        self.offset = offset
        self.blind = blind
        self.optional = list(optional or [])   # OPTIONAL pattern group
        self.unions = [list(u) for u in (unions or [])]  # UNION branches
        # validate var ids HERE: an out-of-range var (< -nvars) would
        # index past the engine's v2c array on the C side
        all_pats = (self.patterns + self.optional
                    + [p for u in self.unions for p in u])
        for (s, p, d, o) in all_pats:
            for v in (s, p, o):
                if isinstance(v, (int, np.integer)) and v < -nvars:
                    raise ValueError(
                        f"variable {v} out of range for nvars={nvars}")
        for v in self.required_vars:
            if v < -nvars or v >= 0:
                raise ValueError(
                    f"required var {v} out of range for nvars={nvars}")

    def to_c(self):
        def mk(pat_list):
            arr = (WkPattern * max(len(pat_list), 1))()
            for i, (s, p, d, o) in enumerate(pat_list):
                arr[i] = WkPattern(s, p, o, d)  # arg order (s,p,d,o) -> struct
            return arr

        pats = mk(self.patterns)
        req = (ctypes.c_int32 * len(self.required_vars))(*self.required_vars)
        plan = WkPlan(ctypes.cast(pats, ctypes.POINTER(WkPattern)),
                      len(self.patterns), self.nvars,
                      ctypes.cast(req, ctypes.POINTER(ctypes.c_int32)),
                      len(self.required_vars), 1 if self.distinct else 0,
                      self.limit, self.offset, 1 if self.blind else 0)
        plan._keepalive = [pats, req]
        if self.optional:
            opt = mk(self.optional)
            plan.opt_patterns = ctypes.cast(opt, ctypes.POINTER(WkPattern))
            plan.nopt = len(self.optional)
            plan._keepalive.append(opt)
        if self.unions:
            flat = [p for u in self.unions for p in u]
            up = mk(flat)
            us = (ctypes.c_int32 * len(self.unions))(*[len(u) for u in self.unions])
            plan.union_pats = ctypes.cast(up, ctypes.POINTER(WkPattern))
            plan.union_sizes = ctypes.cast(us, ctypes.POINTER(ctypes.c_int32))
            plan.nunion = len(self.unions)
            plan._keepalive.extend((up, us))
        return plan


class Store:
    def __init__(self, triples, sid=0, nsrv=1):
        t = np.ascontiguousarray(triples, dtype=np.uint32)
        assert t.ndim == 2 and t.shape[1] == 3
        self._h = _store_build(t.ctypes.data_as(c_u32p), t.shape[0], sid, nsrv)
        if not self._h:
            raise RuntimeError("wk_store_build failed")
        self.sid, self.nsrv = sid, nsrv

    def __del__(self, _free=_store_free):
        # default-arg capture: module globals are already cleared when
        # destructors run at interpreter shutdown
        if getattr(self, "_h", None):
            _free(self._h)
            self._h = None

    def get_triples(self, vid, pid, direction):
        sz = c_u64()
        p = _get_triples(self._h, vid, pid, direction, ctypes.byref(sz))
        if not p or sz.value == 0:
            return np.empty(0, dtype=np.uint32)
        return np.ctypeslib.as_array(p, shape=(sz.value,)).copy()

    def get_index(self, pid, direction):
        sz = c_u64()
        p = _get_index(self._h, pid, direction, ctypes.byref(sz))
        if not p or sz.value == 0:
            return np.empty(0, dtype=np.uint32)
        return np.ctypeslib.as_array(p, shape=(sz.value,)).copy()

    @property
    def num_slots(self):
        return _num_slots(self._h)

    @property
    def num_edges(self):
        return _num_edges(self._h)

    def checksum(self):
        return _checksum(self._h)

    def mem_usage(self):
        """(slots_bytes, edges_bytes, side_index_bytes) — the
        reference's print_mem_usage report (gstore.hpp:1062-1103)."""
        a, b, c = c_u64(), c_u64(), c_u64()
        _mem_usage(self._h, ctypes.byref(a), ctypes.byref(b), ctypes.byref(c))
        return int(a.value), int(b.value), int(c.value)

    def seg_stats(self, pid, direction):
        """(distinct keys, total edges) of the (pid,dir) segment — the
        planner's cost-model inputs (reference stats.hpp)."""
        k, e = c_u64(), c_u64()
        _seg_stats(self._h, pid, direction, ctypes.byref(k), ctypes.byref(e))
        return int(k.value), int(e.value)

    def check(self):
        """gsck-style full integrity scan; returns #violations (0 = ok)."""
        return _store_check(self._h)

    def seg_table(self):
        """Flattened (pid,dir) segment table as an (n,2) uint64 array of
        {bucket_start, num_buckets} — exchanged between ranks for the
        xGMI peer-probe path."""
        n = _seg_table(self._h, None, 0)
        out = np.zeros((max(int(n), 1), 2), dtype=np.uint64)
        if n > 0:
            _seg_table(self._h, out.ctypes.data_as(ctypes.POINTER(c_u64)), n)
        return out[:max(int(n), 0)]


class GpuStore:
    """Device-resident store image shared by multiple engines (one HBM
    upload)."""

    def __init__(self, store, device=0):
        self._store = store
        self._h = _gstore_create(store._h, device)
        if not self._h:
            raise RuntimeError("wk_gpu_store_create failed (no GPU / HBM alloc)")

    def __del__(self, _free=_gstore_destroy):
        if getattr(self, "_h", None):
            _free(self._h)
            self._h = None

    def export_blob(self):
        """HIP-IPC handles + metadata of this rank's HBM store image
        (bytes) for the xGMI peer-probe path."""
        blob = WkPeerBlob()
        rc = _gstore_export(self._h, ctypes.byref(blob))
        if rc != 0:
            raise RuntimeError(f"wk_gpu_store_export rc={rc}")
        return bytes(blob)

    def import_peers(self, blob_bytes_list, seg_tables):
        """Open every rank's store over xGMI (HIP IPC).  blob_bytes_list
        and seg_tables are rank-ordered (this rank's own entries
        included)."""
        n = len(blob_bytes_list)
        blobs = (WkPeerBlob * n)()
        for i, bb in enumerate(blob_bytes_list):
            ctypes.memmove(ctypes.byref(blobs[i]), bb, ctypes.sizeof(WkPeerBlob))
        nseg = len(seg_tables[0])
        flat = np.ascontiguousarray(
            np.stack([np.asarray(t, dtype=np.uint64) for t in seg_tables]))
        rc = _gstore_import(self._h, blobs,
                            flat.ctypes.data_as(ctypes.POINTER(c_u64)),
                            nseg, n)
        if rc != 0:
            raise RuntimeError(f"wk_gpu_store_import_peers rc={rc}")


class Engine:
    """GPU engine bound to one device; requires a GPU (fails loudly)."""

    def __init__(self, store, device=0):
        self._store = store  # keepalive
        if isinstance(store, GpuStore):
            self._h = _eng_create_on(store._h)
        else:
            self._h = _eng_create(store._h, device)
        if not self._h:
            raise RuntimeError(
                "wk_engine_create failed (no MI355X visible, or HBM alloc failed)")

    def submit(self, plan):
        """Enqueue a whole plan asynchronously (harvest with fetch_*)."""
        cplan = plan.to_c()
        rc = _eng_submit(self._h, ctypes.byref(cplan))
        if rc != 0:
            raise RuntimeError(f"submit rc={rc}")
        self._cplan = cplan

    def __del__(self, _free=_eng_destroy):
        if getattr(self, "_h", None):
            _free(self._h)
            self._h = None

    def graph_build(self, plan):
        """Capture the plan's whole launch chain as a hipGraph (one
        warm pass first); returns a graph id for graph_run."""
        cplan = plan.to_c()
        gid = c_i32()
        rc = _eng_graph_build(self._h, ctypes.byref(cplan), ctypes.byref(gid))
        if rc != 0:
            raise RuntimeError(f"graph_build rc={rc}")
        return int(gid.value)

    def graph_run(self, gid):
        """Replay a captured plan (ONE hipGraphLaunch); blind row count.
        Raises OverflowError on capacity overflow — fall back to
        submit()."""
        n = c_i64()
        rc = _eng_graph_run(self._h, gid, ctypes.byref(n))
        if rc == -5:
            raise OverflowError("graph replay overflow: use submit()")
        if rc != 0:
            raise RuntimeError(f"graph_run rc={rc}")
        return int(n.value)

    def graph_build_suite(self, plans):
        """Capture ALL plans back-to-back into ONE hipGraph (pays the
        replay floor once per pass); returns a graph id for
        graph_launch."""
        cplans = (WkPlan * len(plans))()
        keep = []
        for i, p in enumerate(plans):
            cp = p.to_c()
            cplans[i] = cp
            keep.append(cp)
        gid = c_i32()
        rc = _eng_graph_suite(self._h, cplans, len(plans), ctypes.byref(gid))
        if rc != 0:
            raise RuntimeError(f"graph_build_suite rc={rc}")
        self._suite_keep = keep
        return int(gid.value)

    def graph_launch(self, gid):
        """Async replay (no sync); pair with sync()."""
        rc = _eng_graph_launch(self._h, gid)
        if rc != 0:
            raise RuntimeError(f"graph_launch rc={rc}")

    def sync(self):
        """One stream sync; raises OverflowError if any replayed graph
        flagged S_ERR (fall back to per-query submit)."""
        rc = _eng_sync(self._h)
        if rc == -5:
            raise OverflowError("replayed graph overflow")
        if rc != 0:
            raise RuntimeError(f"sync rc={rc}")

    def submit_light_batch(self, subj, pred, dirs, cval):
        """One asynchronous launch for a whole window of light
        (c2u + rdf:type filter) queries — numpy SoA in, counts out via
        wait_light_batch.  Raises ValueError when the store cannot take
        the fast path (caller falls back to per-query submit)."""
        subj = np.ascontiguousarray(subj, dtype=np.int64)
        pred = np.ascontiguousarray(pred, dtype=np.int32)
        dirs = np.ascontiguousarray(dirs, dtype=np.int32)
        cval = np.ascontiguousarray(cval, dtype=np.uint32)
        n = len(subj)
        assert len(pred) == n and len(dirs) == n and len(cval) == n
        rc = _eng_submit_lb(
            self._h,
            subj.ctypes.data_as(ctypes.POINTER(c_i64)),
            pred.ctypes.data_as(ctypes.POINTER(c_i32)),
            dirs.ctypes.data_as(ctypes.POINTER(c_i32)),
            cval.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)), n)
        if rc == -3:
            raise ValueError("store has no single-type index (fast path off)")
        if rc != 0:
            raise RuntimeError(f"submit_light_batch rc={rc}")
        self._lb_n = n

    def submit_plan_batch(self, plan, consts):
        """One launch for a window of same-template whole plans (LDS
        plan interpreter — A4/A6 shapes).  consts replaces the
        template's first-pattern subject per query.  Blind counts via
        wait_light_batch; LP_OVERFLOW entries must be re-run per-query.
        Raises ValueError when the template is not interpretable."""
        consts = np.ascontiguousarray(consts, dtype=np.int64)
        cplan = plan.to_c()
        rc = _eng_submit_pb(self._h, ctypes.byref(cplan),
                            consts.ctypes.data_as(ctypes.POINTER(c_i64)),
                            len(consts))
        if rc == -3:
            raise ValueError("plan not interpretable by the LDS plan batch")
        if rc != 0:
            raise RuntimeError(f"submit_plan_batch rc={rc}")
        self._lb_n = len(consts)
        self._cplan_batch = cplan  # keepalive for pattern array

    def wait_light_batch(self):
        """Block until the window completes; per-query row counts."""
        n = self._lb_n
        counts = np.empty(n, dtype=np.uint64)
        rc = _eng_lb_wait(self._h, counts.ctypes.data_as(ctypes.POINTER(c_u64)), n)
        if rc != 0:
            raise RuntimeError(f"light_batch_wait rc={rc}")
        self._lb_n = None
        return counts

    def run_query_count(self, plan):
        """Blind execution (Result::blind): row count only, no download."""
        import copy
        p2 = copy.copy(plan)
        p2.blind = True
        cplan = p2.to_c()
        res = WkResult()
        rc = _eng_run(self._h, ctypes.byref(cplan), ctypes.byref(res))
        if rc != 0:
            raise RuntimeError(f"wk_engine_run_query(blind) rc={rc}")
        n = res.row_num
        _res_free(ctypes.byref(res))
        return n

    def run_query(self, plan):
        cplan = plan.to_c()
        res = WkResult()
        rc = _eng_run(self._h, ctypes.byref(cplan), ctypes.byref(res))
        if rc != 0:
            raise RuntimeError(f"wk_engine_run_query rc={rc}")
        try:
            if res.row_num and res.col_num and res.table:
                tbl = np.ctypeslib.as_array(
                    res.table, shape=(res.row_num, res.col_num)).copy()
            else:
                tbl = np.empty((0, res.col_num), dtype=np.uint32)
        finally:
            _res_free(ctypes.byref(res))
        return tbl

    # step-level API (multi-GPU driver)
    def begin_query(self, plan):
        cplan = plan.to_c()
        rc = _eng_begin(self._h, ctypes.byref(cplan))
        if rc != 0:
            raise RuntimeError(f"begin_query rc={rc}")
        self._cplan = cplan

    def load_rbuf(self, table, v2c, step):
        t = np.ascontiguousarray(table, dtype=np.uint32)
        v = (ctypes.c_int32 * len(v2c))(*v2c)
        rc = _eng_load(self._h, t.ctypes.data_as(c_u32p),
                       t.shape[0] if t.size else 0,
                       t.shape[1] if t.ndim == 2 else 0, v, step)
        if rc != 0:
            raise RuntimeError(f"load_rbuf rc={rc}")

    def load_rbuf_device(self, dev_ptr, nrows, ncols, v2c, step):
        v = (ctypes.c_int32 * len(v2c))(*v2c)
        rc = _eng_load_dev(self._h, dev_ptr, nrows, ncols, v, step)
        if rc != 0:
            raise RuntimeError(f"load_rbuf_device rc={rc}")

    def execute_one_pattern(self):
        n = c_i64()
        rc = _eng_step(self._h, ctypes.byref(n))
        if rc != 0:
            raise RuntimeError(f"execute_one_pattern rc={rc}")
        return n.value

    def execute_one_pattern_async(self):
        """No host sync — only for steps that cannot overflow (filters:
        outputs <= inputs).  Resolve counts with row_count()."""
        rc = _eng_step(self._h, None)
        if rc != 0:
            raise RuntimeError(f"execute_one_pattern(async) rc={rc}")

    def row_count(self):
        n = c_i64()
        rc = _eng_row_count(self._h, ctypes.byref(n))
        if rc != 0:
            raise RuntimeError(f"row_count rc={rc}")
        return n.value

    def execute_one_pattern_remote(self):
        """Run the current pattern via the xGMI peer-probe path (small
        tables: in-place remote reads instead of an exchange —
        sparql.hpp:802-814).  Raises ValueError for shapes that must
        exchange (caller falls back)."""
        n = c_i64()
        rc = _eng_step_remote(self._h, ctypes.byref(n))
        if rc == -3:  # WK_ERR_PLAN: const-start / per-row index shape
            raise ValueError("pattern shape needs the exchange path")
        if rc != 0:
            raise RuntimeError(f"execute_one_pattern_remote rc={rc}")
        return n.value

    def execute_filter_list(self, sorted_list):
        """Run the current (mid-plan const-/index-start) pattern as a
        membership filter against a caller-supplied sorted list — the
        distributed driver broadcasts the owner rank's edge list first
        (reference: one-sided remote read, gstore.hpp:260-338)."""
        lst = np.ascontiguousarray(sorted_list, dtype=np.uint32)
        n = c_i64()
        rc = _eng_filter_list(self._h, lst.ctypes.data_as(c_u32p),
                              len(lst), ctypes.byref(n))
        if rc != 0:
            raise RuntimeError(f"execute_filter_list rc={rc}")
        return n.value

    @property
    def pattern_step(self):
        return _eng_pattern_step(self._h)

    @property
    def col_num(self):
        return _eng_col_num(self._h)

    def generate_sub_query(self, ndst, dev_out_ptr, cap_rows):
        rows = (c_i64 * ndst)()
        rc = _eng_subq(self._h, ndst, dev_out_ptr, cap_rows, rows)
        if rc != 0:
            raise RuntimeError(f"generate_sub_query rc={rc}")
        return list(rows)

    def fetch_count(self):
        """Blind fetch of the current query (Result::blind): row count."""
        cplan = self._cplan
        old = cplan.blind
        cplan.blind = 1
        res = WkResult()
        rc = _eng_fetch(self._h, ctypes.byref(cplan), ctypes.byref(res))
        cplan.blind = old
        if rc == -5:  # capacity overflow: engine grew, caller resubmits
            return -1
        if rc != 0:
            raise RuntimeError(f"fetch_count rc={rc}")
        n = res.row_num
        _res_free(ctypes.byref(res))
        return n

    def fetch_result(self, plan=None):
        cplan = self._cplan if plan is None else plan.to_c()
        res = WkResult()
        rc = _eng_fetch(self._h, ctypes.byref(cplan), ctypes.byref(res))
        if rc != 0:
            raise RuntimeError(f"fetch_result rc={rc}")
        try:
            if res.row_num and res.col_num:
                tbl = np.ctypeslib.as_array(
                    res.table, shape=(res.row_num, res.col_num)).copy()
            else:
                tbl = np.empty((0, res.col_num), dtype=np.uint32)
        finally:
            _res_free(ctypes.byref(res))
        return tbl

    def fetch_raw(self):
        """Current device table, no final ops (driver/tests)."""
        res = WkResult()
        rc = _eng_fetch_raw(self._h, ctypes.byref(res))
        if rc != 0:
            raise RuntimeError(f"fetch_raw rc={rc}")
        try:
            if res.row_num and res.col_num:
                tbl = np.ctypeslib.as_array(
                    res.table, shape=(res.row_num, res.col_num)).copy()
            else:
                tbl = np.empty((0, res.col_num), dtype=np.uint32)
        finally:
            _res_free(ctypes.byref(res))
        return tbl

    def kernel_stats(self):
        us = (ctypes.c_double * 7)()
        by = (ctypes.c_double * 7)()
        ln = (c_i64 * 7)()
        _kstats(self._h, us, by, ln)
        return {KERNEL_CATS[i]: dict(usec=us[i], bytes=by[i], launches=ln[i])
                for i in range(7)}
