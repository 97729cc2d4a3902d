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


def _cands(store, pat, bound, first, est_rows):
    """Yield (cost, pat, oriented, new_rows) for every legal way to run
    `pat` in the current state.  `first` = nothing planned yet (index /
    const_to_unknown starts are first-pattern-only, query.hpp:660-682)."""
    s0, p0, d0, o0 = pat
    for (a, p, d, b) in ((s0, p0, d0, o0), (o0, p0, d0 ^ 1, s0)):
        if isinstance(p, int) and p < 0:
            # predicate variable: needs const or bound start
            if p in bound:
                continue  # predicate already bound: unsupported
            if a >= 0 and not _is_tpid(a) and first:
                cost = new_rows = float(
                    len(store.get_triples(a, PREDICATE_ID, d)) or 1)
            elif a < 0 and a in bound:
                cost = new_rows = est_rows * 4.0
            else:
                continue
            yield (cost, pat, (a, p, d, b), new_rows)
            continue
        if a >= 0 and _is_tpid(a):
            # index start (type or predicate index)
            if not first or b >= 0:
                continue
            n = len(store.get_index(a, DIR_IN))
            yield (float(max(n, 1)), pat, (a, p, d, b), float(n))
        elif a >= 0:
            if not first and (b < 0 and b not in bound):
                # const_to_unknown is first-pattern-only
                continue
            n = len(store.get_triples(a, p, d))
            if b >= 0 or b in bound:      # membership filter
                yield (float(max(n, 1)) * 0.01 + 1, pat,
                       (a, p, d, b), est_rows)
            else:
                yield (float(max(n, 1)), pat, (a, p, d, b), float(n))
        elif a in bound:
            if b >= 0 or b in bound:      # k2c / k2k: prune
                yield (est_rows * 0.1 + 1, pat, (a, p, d, b),
                       est_rows * 0.5)
            else:                         # k2u: grow by avg degree
                k, e = store.seg_stats(p, d)
                g = est_rows * max((e / k) if k else 0.0, 0.05)
                yield (g + est_rows, pat, (a, p, d, b), g)


def _greedy(store, patterns, first_cand):
    """Complete the order from a forced opener; PlannerError on dead
    end.  Returns (ordered patterns, bound vars, final row estimate)."""
    remaining = list(patterns)
    bound = set()
    out = []
    est_rows = 1.0
    best = first_cand
    while True:
        _, pat, oriented, new_rows = best
        remaining.remove(pat)
        out.append(oriented)
        est_rows = max(new_rows, 1.0)
        for t in (oriented[0], oriented[3], oriented[1]):
            if isinstance(t, int) and t < 0:
                bound.add(t)
        if not remaining:
            return out, bound, est_rows
        best = None
        for p_ in remaining:
            for cand in _cands(store, p_, bound, False, est_rows):
                if best is None or cand[0] < best[0]:
                    best = cand
        if best is None:
            raise PlannerError(f"no startable pattern among {remaining!r}")


def _order_group(store, pats, bound0, est_rows):
    """Orient + greedily order a UNION branch continuing from the main
    BGP's bound set (BGP semantics inside a branch are order-free)."""
    remaining = list(pats)
    bound = set(bound0)
    out = []
    est = est_rows
    while remaining:
        best = None
        for p_ in remaining:
            for cand in _cands(store, p_, bound, False, est):
                if best is None or cand[0] < best[0]:
                    best = cand
        if best is None:
            raise PlannerError(f"no startable pattern in group {remaining!r}")
        _, pat, oriented, new_rows = best
        remaining.remove(pat)
        out.append(oriented)
        est = max(new_rows, 1.0)
        for t in (oriented[0], oriented[3], oriented[1]):
            if isinstance(t, int) and t < 0:
                bound.add(t)
    return out


def _orient_group(store, pats, bound0):
    """Orient an OPTIONAL group's patterns IN TEXTUAL ORDER (the
    matched-flag mechanics are order-sensitive, so only the direction
    may flip, never the sequence)."""
    bound = set(bound0)
    out = []
    for pat in pats:
        best = None
        for cand in _cands(store, pat, bound, False, 1.0):
            if best is None or cand[0] < best[0]:
                best = cand
        if best is None:
            raise PlannerError(f"no orientation for {pat!r} in OPTIONAL")
        oriented = best[2]
        out.append(oriented)
        for t in (oriented[0], oriented[3], oriented[1]):
            if isinstance(t, int) and t < 0:
                bound.add(t)
    return out


def plan_patterns(store, patterns, nvars, required_vars, **plan_kw):
    """patterns: (s, p, d, o) tuples as parsed (textual orientation,
    d=DIR_OUT).  Returns a Plan with a greedy execution order.

    The greedy can dead-end on adversarial inputs: index and
    const_to_unknown starts are first-pattern-only, so opening with the
    wrong one (e.g. a typeof filter flipped into a type-index scan over
    a variable that turns out to be a predicate var) strands the rest.
    Openers are therefore tried in ascending cost with backtracking —
    the first choice is exactly the plain greedy's, so well-formed
    inputs plan identically; a PlannerError only surfaces if EVERY
    opener dead-ends."""
    openers = []
    for pat in patterns:
        openers.extend(_cands(store, pat, set(), True, 1.0))
    openers.sort(key=lambda c: c[0])
    last = None
    for first_cand in openers:
        try:
            out, bound, est = _greedy(store, patterns, first_cand)
            # groups are parsed in textual orientation too: orient
            # union branches (and reorder — BGP semantics) and orient
            # optional patterns (order preserved) from the main BGP's
            # bound set
            kw = dict(plan_kw)
            if kw.get("unions"):
                kw["unions"] = [_order_group(store, br, bound, est)
                                for br in kw["unions"]]
            if kw.get("optional"):
                kw["optional"] = _orient_group(store, kw["optional"], bound)
            return Plan(out, nvars=nvars, required_vars=required_vars, **kw)
        except PlannerError as e:
            last = e
    raise last if last is not None else PlannerError(
        f"no startable pattern among {list(patterns)!r}")


def plan_text(store, text, vocab, **kw):
    """Parse SPARQL text and order it with the greedy planner (for
    queries without an .fmt plan — parser.hpp:284 + planner.hpp:218)."""
    from . import sparql
    p = sparql.parse(text, vocab)
    return plan_patterns(store, p.patterns, p.nvars, p.required_vars,
                         distinct=p.distinct, limit=p.limit, offset=p.offset,
                         optional=p.optional, unions=p.unions)
