"""Text parser + plan application must reproduce the hand-built plans
(and therefore the same results through the oracle)."""
import numpy as np
import pytest

from wukong_amd import queries as Q
from wukong_amd import queries_sparql as QS
from wukong_amd import sparql
from tests.oracle_util import sort_rows


@pytest.mark.parametrize("name", list(QS.TEXT))
def test_parsed_plan_matches_handbuilt(name):
    vocab = sparql.lubm_vocab()
    plan = sparql.parse(QS.TEXT[name], vocab, plan_lines=QS.PLAN[name])
    want = Q.ALL[name]
    # variable numbering may differ (parser numbers by first appearance);
    # compare structurally via a renaming
    assert len(plan.patterns) == len(want.patterns), name
    ren = {}

    def canon(pats, req):
        out = []
        for s, p, d, o in pats:
            out.append((s if s >= 0 else ("v", s), p, d,
                        o if o >= 0 else ("v", o)))
        return out

    # build a var mapping from positionally matching patterns
    for (s1, p1, d1, o1), (s2, p2, d2, o2) in zip(plan.patterns, want.patterns):
        assert p1 == p2 and d1 == d2, (name, (s1, p1, d1, o1), (s2, p2, d2, o2))
        for a, b in ((s1, s2), (o1, o2)):
            if a < 0 or b < 0:
                assert a < 0 and b < 0, name
                assert ren.setdefault(a, b) == b, name
            else:
                assert a == b, name
    assert [ren[v] for v in plan.required_vars] == want.required_vars, name
    assert plan.distinct == want.distinct
    assert plan.limit == want.limit


def test_parsed_results_equal(lubm2):
    from tests.oracle_util import OracleCtx
    ora = OracleCtx(lubm2)
    vocab = sparql.lubm_vocab()
    for name in ("q1", "q5", "q7"):
        p = sparql.parse(QS.TEXT[name], vocab, plan_lines=QS.PLAN[name])
        a = sort_rows(ora.run_query(p))
        b = sort_rows(ora.run_query(Q.ALL[name]))
        # column order follows required_vars in both; var renaming does not
        # change the projected table
        assert np.array_equal(a, b), name


def test_parse_errors():
    vocab = sparql.lubm_vocab()
    with pytest.raises(sparql.ParseError):
        sparql.parse("SELECT ?X WHERE { ?X ub:nope ?Y . }", vocab)
    with pytest.raises(sparql.ParseError):
        sparql.parse("no query here", vocab)


def test_unplanned_textual_order():
    vocab = sparql.lubm_vocab()
    p = sparql.parse(QS.TEXT["q2"], vocab)  # no plan: textual order, OUT
    assert p.patterns[0][2] == sparql.DIR_OUT
    assert len(p.patterns) == 2


def test_parse_optional_union(lubm2):
    """OPTIONAL { } and { } UNION { } parse into Plan.optional/unions
    and execute (oracle) identically to the hand-built groups."""
    import wukong_amd as wk
    from wukong_amd import Plan
    from tests.oracle_util import OracleCtx
    ora = OracleCtx(lubm2)
    vocab = sparql.lubm_vocab()
    text = vocab and (
        "PREFIX rdf: <http://www.w3.org/1999/02/22-rdf-syntax-ns#>\n"
        "PREFIX ub: <http://swat.cse.lehigh.edu/onto/univ-bench.owl#>\n"
        "SELECT ?x ?y WHERE {\n"
        "  ?x rdf:type ub:UndergraduateStudent .\n"
        "  OPTIONAL { ?x ub:advisor ?y . }\n"
        "}")
    from wukong_amd import planner
    store = wk.Store(lubm2)
    plan = planner.plan_text(store, text, vocab)
    assert len(plan.optional) == 1 and not plan.unions
    got = ora.run_query(plan)
    want = ora.run_query(Plan(plan.patterns, plan.nvars, plan.required_vars,
                              optional=plan.optional))
    assert np.array_equal(sort_rows(got), sort_rows(want))
    assert (got == 0xFFFFFFFF).any()  # BLANKs present

    text2 = (
        "PREFIX rdf: <http://www.w3.org/1999/02/22-rdf-syntax-ns#>\n"
        "PREFIX ub: <http://swat.cse.lehigh.edu/onto/univ-bench.owl#>\n"
        "SELECT ?x ?y WHERE {\n"
        "  ?x rdf:type ub:GraduateStudent .\n"
        "  { ?x ub:memberOf ?y . } UNION { ?x ub:undergraduateDegreeFrom ?y . }\n"
        "}")
    plan2 = planner.plan_text(store, text2, vocab)
    assert len(plan2.unions) == 2 and not plan2.optional
    got2 = ora.run_query(plan2)
    assert len(got2) > 0


def test_parser_text_roundtrip_fuzz(lubm2):
    """Random BGPs emitted as SPARQL text, parsed back and planned:
    results must equal the directly-built pattern list planned the
    same way — a parse bug (wrong var ids, wrong URI resolution, a
    dropped triple) shows up as a result mismatch."""
    import random
    import wukong_amd as wk
    from wukong_amd import planner
    from tests.oracle_util import OracleCtx, sort_rows
    import numpy as np

    vocab = sparql.lubm_vocab()
    by_id = {}
    for tok, i in vocab.items():
        by_id.setdefault(i, tok)
    preds = [Q.SUBORG, Q.UGDEGREE, Q.MEMBEROF, Q.WORKSFOR, Q.TEACHEROF,
             Q.ADVISOR, Q.TAKESCOURSE]
    types = [Q.UNIVERSITY, Q.DEPARTMENT, Q.FULLPROF, Q.UGSTUDENT,
             Q.GRADSTUDENT, Q.COURSE]
    store = wk.Store(lubm2)
    oc = OracleCtx(lubm2)
    rng = random.Random(777)
    ran = 0
    for trial in range(30):
        nv = rng.randint(2, 4)
        # textual patterns (all d=OUT; reversed edges emit swapped vars)
        pats = [(-1, Q.TYPE_ID, 1, rng.choice(types))]
        bound = [-1]
        free = [-v for v in range(2, nv + 1)]  # consecutive: -2..-nv
        for _ in range(rng.randint(1, 3)):
            a = rng.choice(bound)
            p = rng.choice(preds)
            if free and rng.random() < 0.7:
                b = free.pop(0)
                bound.append(b)
            else:
                b = rng.choice(bound)
                if b == a:
                    continue
            if rng.random() < 0.5:
                pats.append((a, p, 1, b))
            else:
                pats.append((b, p, 1, a))   # reversed in text
        lines = []
        for (s, p, d, o) in pats:
            st = f"?v{-s - 1}" if s < 0 else by_id[s]
            ot = f"?v{-o - 1}" if o < 0 else by_id[o]
            lines.append(f"  {st} {by_id[p] if p != 1 else 'rdf:type'} {ot} .")
        sel = " ".join(f"?v{-v - 1}" for v in bound)
        text = ("PREFIX ub: <http://swat.cse.lehigh.edu/onto/univ-bench.owl#>\n"
                f"SELECT {sel} WHERE {{\n" + "\n".join(lines) + "\n}")
        try:
            parsed = planner.plan_text(store, text, vocab)
            direct = planner.plan_patterns(store, pats,
                                           max(-v for v in bound), bound)
        except planner.PlannerError:
            continue
        got = oc.run_query(parsed)
        want = oc.run_query(direct)
        assert got.shape == want.shape, (trial, text, got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), (trial, text)
        ran += 1
    assert ran > 20


def test_parser_group_roundtrip_fuzz(lubm2):
    """Same roundtrip net over OPTIONAL { } and { } UNION { } texts:
    group brace extraction and group-var mapping must reproduce the
    directly-built groups (both sides share the planner's orientation,
    so mismatches isolate PARSE bugs)."""
    import random
    import wukong_amd as wk
    from wukong_amd import planner
    from tests.oracle_util import OracleCtx, sort_rows
    import numpy as np

    vocab = sparql.lubm_vocab()
    by_id = {}
    for tok, i in vocab.items():
        by_id.setdefault(i, tok)
    preds = [Q.SUBORG, Q.UGDEGREE, Q.MEMBEROF, Q.WORKSFOR, Q.ADVISOR,
             Q.TAKESCOURSE]
    types = [Q.DEPARTMENT, Q.FULLPROF, Q.UGSTUDENT, Q.GRADSTUDENT]
    store = wk.Store(lubm2)
    oc = OracleCtx(lubm2)
    rng = random.Random(4242)

    def tri_line(s, p, o):
        st = f"?v{-s - 1}" if s < 0 else by_id[s]
        ot = f"?v{-o - 1}" if o < 0 else by_id[o]
        return f"{st} {by_id[p] if p != 1 else 'rdf:type'} {ot} ."

    ran = 0
    for trial in range(30):
        main = [(-1, Q.TYPE_ID, 1, rng.choice(types))]
        gv = -2  # the group-born var
        a, b = rng.choice(preds), rng.choice(preds)
        rev1, rev2 = rng.random() < 0.5, rng.random() < 0.5
        p1 = (gv, a, 1, -1) if rev1 else (-1, a, 1, gv)
        p2 = (gv, b, 1, -1) if rev2 else (-1, b, 1, gv)
        if rng.random() < 0.5:
            text = ("PREFIX ub: <x>\nSELECT ?v0 ?v1 WHERE {\n  "
                    + tri_line(*[main[0][0], main[0][1], main[0][3]])
                    + "\n  { " + tri_line(p1[0], p1[1], p1[3])
                    + " } UNION { " + tri_line(p2[0], p2[1], p2[3]) + " }\n}")
            direct = wk.Plan(main, 2, [-1, gv], unions=[[p1], [p2]])
        else:
            text = ("PREFIX ub: <x>\nSELECT ?v0 ?v1 WHERE {\n  "
                    + tri_line(*[main[0][0], main[0][1], main[0][3]])
                    + "\n  OPTIONAL { " + tri_line(p1[0], p1[1], p1[3])
                    + " }\n}")
            direct = wk.Plan(main, 2, [-1, gv], optional=[p1])
        parsed = planner.plan_text(store, text, vocab)
        want_plan = planner.plan_patterns(store, direct.patterns, 2,
                                          [-1, gv],
                                          optional=direct.optional,
                                          unions=direct.unions)
        got = oc.run_query(parsed)
        want = oc.run_query(want_plan)
        assert got.shape == want.shape, (trial, text, got.shape, want.shape)
        assert np.array_equal(sort_rows(got), sort_rows(want)), (trial, text)
        ran += 1
    assert ran > 25
