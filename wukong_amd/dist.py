"""Distributed (multi-GPU) query driver: one process per GPU, store
partitioned by vid % world exactly as the reference partitions by server
(core/loader/base_loader.hpp:284, gstore.hpp:1050).

Per pattern step whose start variable is not the one the rows are local
by, the driver decides on the GLOBAL row total (need_fork_join,
sparql.hpp:802-814): at or above the threshold the binding table is
split by `vid % world` (generate_sub_query, sparql.hpp:746-799) and
exchanged with ONE all-to-allv — RCCL over xGMI on the nccl backend,
an object gather on gloo (CPU tests); below it the rows stay put and
k_peer_step probes the OWNER rank's store in place through HIP-IPC
xGMI mappings (the one-sided-RDMA analog, gstore.hpp:260-338).
Mid-plan const-/index-start filters broadcast the owner's edge list;
filter steps between exchanges launch asynchronously; the final
DISTINCT/OFFSET/LIMIT run once after the rank merge.  UNION branches
re-enter the same step loop as engine sub-query continuations;
OPTIONAL groups run as a host-side matched-flag restatement over
owner-exchanged rows; VERSATILE steps exchange onto the owner-local
vp lists (details per method below).  Every rank
executes every step on its local store: probes of non-local keys miss
by construction, so no owner special-casing is needed (index/const
starts are naturally local, dispatch semantics of
sparql.hpp:1064-1111).
"""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

# the reference's fork-join gate (Global::rdma_threshold=300,
# need_fork_join sparql.hpp:802-814): tables below this read the owner
# rank's store in place instead of exchanging
RDMA_THRESHOLD = int(os.environ.get("WK_RDMA_THRESHOLD", "300"))


def _is_tpid(x):
    return 1 < x < (1 << 17)


def init_peers(gstore, store):
    """Exchange HIP-IPC store handles + segment tables between ranks and
    open xGMI peer mappings — the one-sided remote-read analog
    (gstore.hpp:260-338).  All ranks must agree on the outcome (the
    remote/exchange decision gates a collective), so the success bit is
    min-reduced; on any failure every rank keeps the exchange-only path."""
    if not dist.is_initialized() or dist.get_world_size() < 2:
        return False
    world = dist.get_world_size()
    ok = 1
    try:
        blob = gstore.export_blob()
        segs = store.seg_table()
        gathered = [None] * world
        dist.all_gather_object(gathered, (blob, segs))
        nseg = len(gathered[0][1])
        if any(len(g[1]) != nseg for g in gathered):
            ok = 0
        else:
            gstore.import_peers([g[0] for g in gathered],
                                [g[1] for g in gathered])
    except Exception as ex:
        print(f"[dist] peer import unavailable: {ex}", file=sys.stderr)
        ok = 0
    dev = "cuda" if dist.get_backend() == "nccl" else "cpu"
    t = torch.tensor([ok], dtype=torch.int64, device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    ready = bool(int(t.item()))
    gstore._peers_ready = ready
    return ready


def plan_v2c_states(plan):
    """v2c map AFTER each step (pure function of the plan — mirrors the
    column-assignment rules of the operators, query.hpp:352-374)."""
    v2c = [-1] * plan.nvars
    col = 0
    states = []
    for i, (s, p, d, o) in enumerate(plan.patterns):
        def known(v):
            return v < 0 and v2c[-(v + 1)] >= 0
        if p < 0:
            # VERSATILE: the predicate var binds a new column, an
            # unknown object var the one after (sparql.hpp:556-744
            # *_unknown_unknown / *_unknown_const column order)
            v2c[-(p + 1)] = col
            if o < 0:
                v2c[-(o + 1)] = col + 1
                col += 2
            else:
                col += 1
        elif i == 0 and s >= 0 and _is_tpid(s):
            v2c[-(o + 1)] = 0
            col = 1
        elif s >= 0 and o < 0 and not known(o):
            v2c[-(o + 1)] = col
            col += 1
        elif s < 0 and o < 0 and not known(o):
            v2c[-(o + 1)] = col
            col += 1
        states.append((list(v2c), col))
    return states


class DistQuery:
    """Runs one plan across all ranks.  executor: step-level interface
    (wukong_amd.Engine on GPU; tests use the oracle executor)."""

    def __init__(self, executor, plan, rank, world, device=None,
                 threshold=RDMA_THRESHOLD):
        self.ex = executor
        self.plan = plan
        self.rank = rank
        self.world = world
        self.device = device
        self.threshold = threshold
        self.states = plan_v2c_states(plan)

    def _global_rows(self):
        """Total rows across ranks (the remote/exchange decision must be
        uniform — the exchange is a collective)."""
        dev = ("cuda" if dist.is_initialized()
               and dist.get_backend() == "nccl" else "cpu")
        t = torch.tensor([int(self.ex.rows())], dtype=torch.int64, device=dev)
        dist.all_reduce(t)
        return int(t.item())

    def _try_remote(self, i, pat, states):
        """Sub-threshold tables probe the owner rank's store in place
        (xGMI peer mappings; oracle peers in CPU tests) instead of
        exchanging — need_fork_join, sparql.hpp:802-814."""
        if self.threshold <= 0 or not self.ex.supports_remote():
            return False
        if self._global_rows() >= self.threshold:
            return False
        v2c_prev, _ = states[i - 1]
        v2c_next, _ = states[i]
        try:
            self.ex.step_remote(i, pat, v2c_prev, v2c_next)
            return True
        except ValueError:
            return False  # shape needs the exchange (same on every rank)

    # ---- exchanges ----
    def _exchange_cpu(self, table, col):
        """gloo path: split rows by value % world, gather objects."""
        chunks = [table[table[:, col] % self.world == d] for d in range(self.world)]
        gathered = [None] * self.world
        dist.all_gather_object(gathered, chunks)
        mine = [g[self.rank] for g in gathered]
        return np.concatenate(mine, axis=0) if mine else table[:0]

    def _exchange_nccl(self, engine, ncols, col):
        """RCCL all-to-allv of row chunks (replaces the reference's
        GPUDirect-RDMA chunk WRITE, rdma_adaptor.hpp:339-364)."""
        # pack rows into per-destination contiguous chunks on device
        # (ex.rows() resolves any async filter steps first)
        nrows = max(self.ex.rows(), 1)
        buf = torch.empty(nrows * ncols + 1, dtype=torch.int32,
                          device=self.device)
        sizes = engine.generate_sub_query(self.world, buf.data_ptr(), nrows)
        send_rows = torch.tensor(sizes, dtype=torch.int64, device=self.device)
        recv_rows = torch.empty_like(send_rows)
        dist.all_to_all_single(recv_rows, send_rows)
        in_splits = [int(s) * ncols for s in sizes]
        out_list = [int(x) * ncols for x in recv_rows.tolist()]
        # global-empty guard (e.g. a query emptied on every rank): all
        # ranks must agree before skipping the collective
        tot = torch.tensor([sum(in_splits)], dtype=torch.int64, device=self.device)
        dist.all_reduce(tot)
        if int(tot.item()) == 0:
            return torch.empty(1, dtype=torch.int32, device=self.device), 0
        recv = torch.empty(max(sum(out_list), 1), dtype=torch.int32,
                           device=self.device)
        dist.all_to_all_single(recv[:sum(out_list)],
                               buf[:sum(in_splits)], out_list, in_splits)
        return recv, sum(out_list) // ncols if ncols else 0

    def run(self):
        plan = self.plan
        local_var = self._run_steps(plan, self.states, 0, None)
        unions = getattr(plan, "unions", [])
        if unions:
            self._run_unions(unions, local_var)
        optional = getattr(plan, "optional", [])
        if optional:
            self._run_optional(optional)

    def _run_unions(self, unions, local_var):
        """UNION branches, distributed: each branch inherits the merged
        main-BGP table (snapshot/restore) and runs through the SAME
        step machinery — exchanges, remote reads and broadcast filters
        included — then the branch outputs concatenate
        (execute_sparql_query, sparql.hpp:1564-1601; merge semantics
        rmap.hpp:57-87, first branch's column layout as in the oracle).
        Final ops run once after the rank merge, in gather_result."""
        from . import Plan
        plan = self.plan
        nmain = len(plan.patterns)
        base_v2c, _ = self.states[-1]
        self.ex.rows()  # resolve any async filter step before snapshot
        snapshot = self.ex.table()  # host copy of this rank's shard
        parts, final_v2c = [], None
        for branch in unions:
            nv = plan.nvars
            for (s, p, d, o) in branch:
                for v in (s, o):
                    if v < 0:
                        nv = max(nv, -v)
            plan_b = Plan(list(plan.patterns) + list(branch), nv,
                          plan.required_vars)
            states_b = plan_v2c_states(plan_b)
            # pad the inherited v2c to the branch's nvars (var -k lives
            # at index k-1, so new branch vars extend at the tail)
            v2c_b = list(base_v2c) + [-1] * (nv - plan.nvars)
            self.ex.rebind(plan_b, snapshot, v2c_b, nmain)
            self._run_steps(plan_b, states_b, nmain, local_var)
            self.ex.rows()  # resolve a trailing async filter step
            parts.append(self.ex.table())
            if final_v2c is None:
                final_v2c = states_b[-1][0]
        self._union_parts = parts
        self._union_v2c = final_v2c

    BLANK = 0xFFFFFFFF

    def _exchange_rows_cpu(self, table, matched, col):
        """Host-side exchange of (rows, matched flags) by owner of
        row[col]; rows whose key is BLANK stay put (they never probe)."""
        w = self.world
        key_ok = table[:, col] != np.uint32(self.BLANK)
        stay_t, stay_m = table[~key_ok], matched[~key_ok]
        chunks = []
        for dst in range(w):
            sel = key_ok & (table[:, col] % w == dst)
            chunks.append((table[sel], matched[sel]))
        gathered = [None] * w
        dist.all_gather_object(gathered, chunks)
        ts = [stay_t] + [g[self.rank][0] for g in gathered]
        ms = [stay_m] + [g[self.rank][1] for g in gathered]
        return (np.concatenate(ts, axis=0), np.concatenate(ms))

    def _run_optional(self, opt_pats):
        """OPTIONAL group, distributed: a HOST-side restatement of the
        reference's matched-flag mechanics (opt_mode row semantics,
        sparql.hpp:100-170,316-375,416-549 under :1603-1662) — the
        post-pattern passes run on the pruned table, so
        per-row work is host-side; owner-correct probes come from
        exchanging rows to the key's owner rank before each step and
        broadcasting const-start edge lists.  Columns born inside the
        group (opt_mask) blank out when a later step unmatches the row;
        a matched degree-0 row keeps its flag (the reference's
        left-join quirk)."""
        ex, plan = self.ex, self.plan
        if getattr(self, "_union_parts", None) is not None:
            parts = [t for t in self._union_parts if t.size]
            T = (np.concatenate(parts, axis=0) if parts
                 else self._union_parts[0])
            v2c = list(self._union_v2c)
        else:
            ex.rows()
            T = np.asarray(ex.table())
            v2c = list(self.states[-1][0])
        ncols = T.shape[1] if T.ndim == 2 else 0
        T = T.reshape(-1, max(ncols, 1)).astype(np.uint32, copy=True)
        matched = np.ones(len(T), dtype=bool)
        opt_cols = []
        B = np.uint32(self.BLANK)

        def blank_rows(mask):
            for c in opt_cols:
                T[mask, c] = B

        for (s, p, d, o) in opt_pats:
            if p < 1 or (p == 1 and d == 0):
                # predicate variables and type-member expansion (the
                # [0|tid|IN] index is spread over every rank) don't fit
                # the row-owner exchange; no reference plan uses them
                # inside OPTIONAL
                raise ValueError(
                    "unsupported pattern shape inside a distributed "
                    "OPTIONAL group")
            if s >= 0:
                # const_to_known under OPTIONAL: broadcast the owner's
                # edge list, then blank+unmatch rows outside it
                local = (ex.get_index(s, d) if _is_tpid(s)
                         else ex.get_triples(s, p, d))
                gathered = [None] * self.world
                dist.all_gather_object(
                    gathered, np.asarray(local, dtype=np.uint32))
                merged = np.unique(np.concatenate(gathered))
                col = v2c[-(o + 1)]
                ok = (T[:, col] != B) & np.isin(T[:, col], merged)
                drop = ~ok
                blank_rows(drop & matched)
                matched &= ok
                continue
            col = v2c[-(s + 1)]
            T, matched = self._exchange_rows_cpu(T, matched, col)
            ostat = 2 if o >= 0 else (1 if v2c[-(o + 1)] >= 0 else 0)
            memo = {}  # per-key edge memo (the dup memo, sparql.hpp:322-345)

            def edges_of(cur):
                e_ = memo.get(cur)
                if e_ is None:
                    e_ = ex.get_triples(cur, p, d)
                    memo[cur] = e_
                return e_

            if ostat == 0:
                # k2u left join, vectorized per unique probe key (row
                # order is free — the result is a multiset)
                cur_vals = T[:, col]
                active = matched & (cur_vals != B)
                uniq, inv = np.unique(cur_vals, return_inverse=True)
                parts_t, parts_c, parts_f = [], [], []
                ext_rows = np.zeros(len(T), dtype=bool)
                for u_i, u in enumerate(uniq):
                    if u == B:
                        continue
                    edges = np.asarray(edges_of(int(u)), dtype=np.uint32)
                    if len(edges) == 0:
                        continue
                    sel = active & (inv == u_i)
                    n = int(sel.sum())
                    if n == 0:
                        continue
                    ext_rows |= sel
                    parts_t.append(np.repeat(T[sel], len(edges), axis=0))
                    parts_c.append(np.tile(edges, n))
                    parts_f.append(np.ones(n * len(edges), dtype=bool))
                keep = ~ext_rows  # unmatched/BLANK keep flag; deg-0
                if keep.any():    # matched keeps flag True
                    parts_t.append(T[keep])
                    parts_c.append(np.full(int(keep.sum()), B, np.uint32))
                    parts_f.append(matched[keep] | active[keep])
                if parts_t:
                    T = np.column_stack(
                        [np.concatenate(parts_t, axis=0),
                         np.concatenate(parts_c)]).astype(np.uint32)
                    matched = np.concatenate(parts_f)
                else:
                    T = np.empty((0, T.shape[1] + 1), dtype=np.uint32)
                    matched = np.zeros(0, dtype=bool)
                v2c[-(o + 1)] = ncols
                opt_cols.append(ncols)
                ncols += 1
            else:
                # k2k / k2c: vectorized per unique probe key
                cur_vals = T[:, col]
                tgts = (np.full(len(T), o, dtype=np.uint32) if ostat == 2
                        else T[:, v2c[-(o + 1)]])
                ok = np.zeros(len(T), dtype=bool)
                uniq, inv = np.unique(cur_vals, return_inverse=True)
                for u_i, u in enumerate(uniq):
                    if u == B:
                        continue
                    edges = edges_of(int(u))
                    sel = inv == u_i
                    ok[sel] = (tgts[sel] != B) & np.isin(tgts[sel], edges)
                blank_rows(~ok & matched)
                matched &= ok
        self._opt_table = T
        self._opt_v2c = v2c

    def _run_steps(self, plan, states, start, local_var):
        ex = self.ex
        nccl = dist.is_initialized() and dist.get_backend() == "nccl"
        for i in range(start, len(plan.patterns)):
            s, p, d, o = plan.patterns[i]
            if i > 0 and s >= 0 and p < 0:
                # mid-plan const predicate-variable: the engine (and the
                # reference's dispatcher) only accept const_unknown_* as
                # the FIRST pattern (sparql.hpp:719) — it regenerates
                # the table from the constant, discarding prior bindings
                raise ValueError(
                    "const predicate-variable pattern must be first")
            if i > 0 and s >= 0:
                # mid-plan const-/index-start membership filter: the
                # start's edge list lives only on its owner rank
                # (`s % world`), so filtering against the local store
                # would drop every row elsewhere.  Broadcast the owner's
                # list and filter once against the merged set — the
                # reference reads it in place over one-sided RDMA
                # (gstore.hpp:260-338).
                local = (ex.get_index(s, d) if _is_tpid(s)
                         else ex.get_triples(s, p, d))
                gathered = [None] * self.world
                dist.all_gather_object(
                    gathered, np.asarray(local, dtype=np.uint32))
                merged = np.unique(np.concatenate(gathered))  # sorted
                v2c_prev, _ = states[i - 1]
                ex.filter_with_list(merged, i, v2c_prev,
                                    v2c_prev[-(o + 1)])
                continue
            if i > 0 and s < 0 and local_var != s:
                if self._try_remote(i, (s, p, d, o), states):
                    continue  # rows stayed put; owner stores were probed
                # fork-join exchange (need_fork_join, sparql.hpp:802-814)
                v2c, ncols = states[i - 1]
                col = v2c[-(s + 1)]
                if nccl:
                    recv, nrows = self._exchange_nccl(ex.engine, ncols, col)
                    ex.engine.load_rbuf_device(recv.data_ptr(), nrows, ncols,
                                               v2c, i)
                    del recv
                else:
                    table = ex.table()
                    merged = self._exchange_cpu(table, col)
                    ex.load(merged, v2c, i)
                local_var = s
            ex.step()
            if i == 0:
                # i2u sets local_var=end (sparql.hpp:230); c2u leaves it
                # unset (rows are edge values, arbitrary ranks)
                local_var = o if (s >= 0 and _is_tpid(s)) else None
        return local_var

    def gather_result(self):
        """Final table: gather rank results (rmap merge semantics,
        rmap.hpp:57-87), with the reference's final ops run ONCE after
        the merge (final_process, sparql.hpp:1424-1551) — per-rank
        DISTINCT would keep cross-rank duplicates; per-rank LIMIT/OFFSET
        would return world*limit / drop world*offset rows."""
        p = self.plan
        if getattr(self, "_opt_table", None) is not None:
            # optional ran host-side (after any union merge): the final
            # table and its v2c live on the driver
            gathered = [None] * self.world
            dist.all_gather_object(gathered, self._opt_table)
            full = np.concatenate(
                [g for g in gathered if g.size] or [self._opt_table], axis=0)
            return final_process(full, self._opt_v2c, p)
        if getattr(self, "_union_parts", None) is not None:
            # union plans: branch outputs already concatenate host-side
            # (first branch's column layout, as in the oracle); every
            # final op — including plain projection — runs on the merge
            raw = (np.concatenate([t for t in self._union_parts if t.size],
                                  axis=0)
                   if any(t.size for t in self._union_parts)
                   else self._union_parts[0])
            gathered = [None] * self.world
            dist.all_gather_object(gathered, raw)
            full = np.concatenate([g for g in gathered if g.size] or [raw],
                                  axis=0)
            return final_process(full, self._union_v2c, p)
        if not (p.distinct or p.limit >= 0 or p.offset > 0):
            part = self.ex.finalize()  # projection only — order-free
            gathered = [None] * self.world
            dist.all_gather_object(gathered, part)
            return np.concatenate([g for g in gathered if g.size] or [part],
                                  axis=0)
        raw = self.ex.table()
        gathered = [None] * self.world
        dist.all_gather_object(gathered, raw)
        full = np.concatenate([g for g in gathered if g.size] or [raw], axis=0)
        v2c, _ = self.states[-1]
        return final_process(full, v2c, p)


def final_process(table, v2c, plan):
    """The reference's final ops (final_process, sparql.hpp:1424-1551):
    DISTINCT = full-row sort + adjacent equal-on-required-vars removal,
    then OFFSET, LIMIT, projection to the required-var columns."""
    cols = [v2c[-(v + 1)] for v in plan.required_vars]
    t = np.asarray(table)
    if plan.distinct and len(t):
        t = t[np.lexsort(t.T[::-1])]
        reqt = t[:, cols]
        keep = np.ones(len(t), dtype=bool)
        keep[1:] = (reqt[1:] != reqt[:-1]).any(axis=1)
        t = t[keep]
    if plan.offset > 0:
        t = t[plan.offset:]
    if plan.limit >= 0:
        t = t[:plan.limit]
    return np.ascontiguousarray(t[:, cols])


class GpuExecutor:
    """wukong_amd.Engine adapter for DistQuery (nccl path keeps tables on
    device; gloo path round-trips through host for the CPU tests)."""

    def __init__(self, wk_engine, plan):
        self.engine = wk_engine
        self.plan = plan
        self.engine.begin_query(plan)
        self.engine.last_rows = 0
        self._stale = False  # async filter launched; count not synced
        self._host_table = None
        # host store (edge-list reads for the const-start broadcast)
        s = wk_engine._store
        self.store = s if hasattr(s, "get_triples") else s._store

    def get_triples(self, vid, pid, d):
        return self.store.get_triples(vid, pid, d)

    def get_index(self, pid, d):
        return self.store.get_index(pid, d)

    def filter_with_list(self, sorted_list, step, v2c, col):
        n = self.engine.execute_filter_list(sorted_list)  # syncs inside
        self.engine.last_rows = n
        self._stale = False
        return n

    def rows(self):
        if self._stale:
            self.engine.last_rows = self.engine.row_count()
            self._stale = False
        return self.engine.last_rows

    def supports_remote(self):
        return bool(getattr(self.engine._store, "_peers_ready", False))

    def step_remote(self, i, pat, v2c_prev, v2c_next):
        n = self.engine.execute_one_pattern_remote()
        self.engine.last_rows = n
        self._stale = False
        return n

    def load(self, table, v2c, step):
        self.engine.load_rbuf(table, v2c, step)
        self.engine.last_rows = len(table)
        self._stale = False

    def rebind(self, plan_b, table, v2c, step):
        """Restart the engine on a continuation plan with an inherited
        table — the UNION-branch snapshot/restore (run_union semantics,
        sparql.hpp:1564-1601, via the sub-query continuation interface
        begin_query + load_rbuf at step>0)."""
        self.plan = plan_b
        if hasattr(self, "_states"):
            del self._states
        self.engine.begin_query(plan_b)
        self.load(table, v2c, step)

    def step(self):
        # filter steps (known/const end: outputs <= inputs, no overflow
        # possible) launch WITHOUT a host sync — the count resolves
        # lazily at the next exchange point or fetch, so the per-rank
        # chain between exchanges stays on-stream (VERDICT round-1
        # item 8's goal without a graph capture)
        i = self.engine.pattern_step
        pats = self.plan.patterns
        if 0 < i < len(pats):
            if not hasattr(self, "_states"):
                self._states = plan_v2c_states(self.plan)
            s, p, d, o = pats[i]
            known = o >= 0 or self._states[i - 1][0][-(o + 1)] >= 0
            # p<0 (VERSATILE) can GROW the table even with a known end
            # (one row per matching predicate) — never overflow-safe
            if s < 0 and p >= 1 and known:
                self.engine.execute_one_pattern_async()
                self._stale = True
                return None
        n = self.engine.execute_one_pattern()
        self.engine.last_rows = n
        self._stale = False
        return n

    @property
    def col_num(self):
        return self.engine.col_num

    def table(self):
        # host copy of the current device table (gloo test path)
        return self.engine.fetch_raw()

    def finalize(self):
        return self.engine.fetch_result()
