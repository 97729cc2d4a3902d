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
            if b < 0 and b in bound:
                # known_unknown_known: absent in the reference and
                # rejected by the engine (ostat==1) — ordering a vu
                # AFTER its object var binds would change semantics
                continue
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


def _search(store, patterns, bound0, first, est0, budget=20000):
    """Cost-ordered depth-first search over execution orders: at each
    level try the candidates cheapest-first and backtrack on dead ends.
    The first complete leaf IS the plain greedy's order, so well-formed
    inputs cost one pass; adversarial orders (e.g. a flipped k2u that
    would bind a variable another pattern needs as its PREDICATE var)
    backtrack instead of failing.  Returns (out, bound, est_rows)."""
    picked = []  # (oriented, new_rows) along the accepted path
    state = [0]  # node budget box

    def dfs(remaining, bound, est, is_first):
        if not remaining:
            return True
        cands = []
        for p_ in remaining:
            cands.extend(_cands(store, p_, bound, is_first, est))
        cands.sort(key=lambda c: c[0])
        for cost, pat, oriented, new_rows in cands:
            state[0] += 1
            if state[0] > budget:
                raise PlannerError("planner search budget exceeded")
            nb = set(bound)
            for t in (oriented[0], oriented[3], oriented[1]):
                if isinstance(t, int) and t < 0:
                    nb.add(t)
            picked.append((oriented, new_rows))
            if dfs([p_ for p_ in remaining if p_ is not pat], nb,
                   max(new_rows, 1.0), False):
                return True
            picked.pop()
        return False

    if not dfs(list(patterns), set(bound0), est0, first):
        raise PlannerError(f"no startable order for {list(patterns)!r}")
    out = [o for o, _ in picked]
    bound = set(bound0)
    for oriented in out:
        for t in (oriented[0], oriented[3], oriented[1]):
            if isinstance(t, int) and t < 0:
                bound.add(t)
    est = max(picked[-1][1], 1.0) if picked else est0
    return out, bound, est


def _order_group(store, pats, bound0, est_rows):
    """Orient + order a UNION branch continuing from the main BGP's
    bound set (BGP semantics inside a branch are order-free); same
    backtracking search as the main BGP."""
    out, _, _ = _search(store, pats, bound0, False, est_rows)
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

    A plain greedy can dead-end on adversarial inputs: index and
    const_to_unknown starts are first-pattern-only, and a flipped k2u
    can bind a variable another pattern needs as its PREDICATE var —
    either way the rest of the BGP strands even though a valid order
    exists.  `_search` therefore explores candidates cheapest-first
    with full backtracking; its first complete leaf IS the greedy
    order, so well-formed inputs plan identically in one pass."""
    out, bound, est = _search(store, patterns, set(), True, 1.0)
    # groups are parsed in textual orientation too: orient union
    # branches (and reorder — BGP semantics) and orient optional
    # patterns (order preserved) from the main BGP's bound set
    kw = dict(plan_kw)
    if kw.get("unions"):
        kw["unions"] = [_order_group(store, br, bound, est)
                        for br in kw["unions"]]
    if kw.get("optional"):
        kw["optional"] = _orient_group(store, kw["optional"], bound)
    return Plan(out, nvars=nvars, required_vars=required_vars, **kw)


def plan_text(store, text, vocab, **kw):
    """Parse SPARQL text and order it with the greedy planner (for
    queries without an .fmt plan — parser.hpp:284 + planner.hpp:218)."""
    from . import sparql
    p = sparql.parse(text, vocab)
    return plan_patterns(store, p.patterns, p.nvars, p.required_vars,
                         distinct=p.distinct, limit=p.limit, offset=p.offset,
                         optional=p.optional, unions=p.unions)
