"""Greedy cost-based query planner (control plane, above the hot path).

Minimal port of the reference planner's job (core/optimizer/planner.hpp
:218 `generate_for_patterns` + core/optimizer/stats.hpp type-centric
statistics): order and orient a parsed BGP so every pattern starts from
a constant, a type/predicate index, or an already-bound variable, and
greedily minimise the estimated intermediate-table growth.

Cost model inputs:
  - exact first-hop fan-outs for constant starts (one host probe,
    `store.get_triples` — cheaper and tighter than the reference's
    sampled stats);
  - per-(pid,dir) average degree = edges/keys from
    `wk_store_seg_stats` (the reference's stats aggregates);
  - type/predicate index sizes via `store.get_index`.

The output is a `Plan` whose pattern order the engine's dispatcher
accepts (known-start invariant, sparql.hpp:1016-1058).
"""
from . import Plan

PREDICATE_ID, TYPE_ID = 0, 1
DIR_IN, DIR_OUT = 0, 1


def _is_tpid(x):
    return 0 < x < (1 << 17)


class PlannerError(ValueError):
    pass


def plan_patterns(store, patterns, nvars, required_vars, **plan_kw):
    """patterns: (s, p, d, o) tuples as parsed (textual orientation,
    d=DIR_OUT).  Returns a Plan with a greedy execution order."""
    remaining = list(patterns)
    bound = set()
    out = []
    est_rows = 1.0

    def avg_deg(p, d):
        k, e = store.seg_stats(p, d)
        return (e / k) if k else 0.0

    def candidates(pat):
        s, p, d, o = pat
        # both orientations of the stored direction: (start,pred,dir,end)
        yield (s, p, d, o)
        yield (o, p, d ^ 1, s)

    while remaining:
        best = None  # (cost, pat, oriented, new_rows)
        for pat in remaining:
            for (a, p, d, b) in candidates(pat):
                if isinstance(p, int) and p < 0:
                    # predicate variable: needs const or bound start
                    if p in bound:
                        continue  # predicate already bound: unsupported
                    if a >= 0 and not _is_tpid(a) and not out:
                        cost = new_rows = float(
                            len(store.get_triples(a, PREDICATE_ID, d)) or 1)
                    elif a < 0 and a in bound:
                        cost = new_rows = est_rows * 4.0
                    else:
                        continue
                    cand = (cost, pat, (a, p, d, b), new_rows)
                    if best is None or cand[0] < best[0]:
                        best = cand
                    continue
                if a >= 0 and _is_tpid(a):
                    # index start (type or predicate index): first pattern
                    # only (query.hpp:660-682)
                    if out or b >= 0:
                        continue
                    n = len(store.get_index(a, DIR_IN))
                    cand = (float(max(n, 1)), pat, (a, p, d, b), float(n))
                elif a >= 0:
                    if out and (b < 0 and b not in bound):
                        # const_to_unknown is first-pattern-only
                        continue
                    n = len(store.get_triples(a, p, d))
                    if b >= 0 or b in bound:      # membership filter
                        cand = (float(max(n, 1)) * 0.01 + 1, pat,
                                (a, p, d, b), est_rows)
                    else:
                        cand = (float(max(n, 1)), pat, (a, p, d, b), float(n))
                elif a in bound:
                    if b >= 0 or b in bound:      # k2c / k2k: prune
                        cand = (est_rows * 0.1 + 1, pat, (a, p, d, b),
                                est_rows * 0.5)
                    else:                         # k2u: grow by avg degree
                        g = est_rows * max(avg_deg(p, d), 0.05)
                        cand = (g + est_rows, pat, (a, p, d, b), g)
                else:
                    continue
                if best is None or cand[0] < best[0]:
                    best = cand
        if best is None:
            # special case: ?X rdf:type T as the opener -> type-index
            for pat in remaining:
                s, p, d, o = pat
                if (not out and p == TYPE_ID and d == DIR_OUT and s < 0
                        and o >= 0 and _is_tpid(o)):
                    n = len(store.get_index(o, DIR_IN))
                    best = (float(n), pat, (o, TYPE_ID, DIR_IN, s), float(n))
                    break
        if best is None:
            raise PlannerError(f"no startable pattern among {remaining!r}")
        _, pat, oriented, new_rows = best
        remaining.remove(pat)
        out.append(oriented)
        est_rows = max(new_rows, 1.0)
        for t in (oriented[0], oriented[3], oriented[1]):
            if isinstance(t, int) and t < 0:
                bound.add(t)

    return Plan(out, nvars=nvars, required_vars=required_vars, **plan_kw)


def plan_text(store, text, vocab, **kw):
    """Parse SPARQL text and order it with the greedy planner (for
    queries without an .fmt plan — parser.hpp:284 + planner.hpp:218)."""
    from . import sparql
    p = sparql.parse(text, vocab)
    return plan_patterns(store, p.patterns, p.nvars, p.required_vars,
                         distinct=p.distinct, limit=p.limit, offset=p.offset,
                         optional=p.optional, unions=p.unions)
