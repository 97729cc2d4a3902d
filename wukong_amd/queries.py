"""LUBM Q1-Q7 plans, restated from the reference's OSDI16 benchmark
configuration (scripts/sparql_query/lubm/basic/lubm_q{1..7} +
osdi16_plan/*.fmt).  Plan-line semantics: `N >` keeps pattern N as
(s, p, OUT, o); `N <` reverses it to (o, p, IN, s); `N <<` starts from
the predicate index (pid, PREDICATE_ID, IN, ?s)
(core/parser.hpp plan format, documented in the .fmt headers).

Pattern tuples here are (start, predicate, direction, end); vars are
negative ids.  Schema ids match wukong_amd/csrc/lubm_gen.cpp.
"""
from . import Plan, DIR_IN, DIR_OUT

# schema ids (lubm_gen.cpp enum; fixed enumeration, ids stable)
PREDICATE_ID, TYPE_ID = 0, 1
SUBORG, UGDEGREE, MEMBEROF, WORKSFOR, TEACHEROF, ADVISOR, TAKESCOURSE = range(2, 9)
NAME, EMAIL, TELEPHONE, HEADOF, DOCDEGREE = range(9, 14)
(UNIVERSITY, DEPARTMENT, FULLPROF, ASSOCPROF, ASSTPROF, LECTURER,
 UGSTUDENT, GRADSTUDENT, COURSE, GRADCOURSE, RESEARCHGROUP) = range(14, 25)
PUBAUTHOR, PUBLICATION = 25, 26

# deterministic generator constants (lubm_gen.cpp allocation order):
UNIV0 = 1 << 17            # university 0 entity
DEPT0_UNIV0 = UNIV0 + 2    # univ, univ-name-literal, then first dept

X, Y, Z, Y1, Y2, Y3 = -1, -2, -3, -2, -3, -4

# Q1 — osdi16_plan/lubm_q1.fmt: 3< 4> 2> 1> 5< 6>
Q1 = Plan([
    (GRADSTUDENT, TYPE_ID, DIR_IN, X),      # ?X type GraduateStudent (idx)
    (X, MEMBEROF, DIR_OUT, Z),
    (X, UGDEGREE, DIR_OUT, Y),
    (Y, TYPE_ID, DIR_OUT, UNIVERSITY),
    (Y, SUBORG, DIR_IN, Z),                 # ?Z subOrgOf ?Y reversed -> k2k
    (Z, TYPE_ID, DIR_OUT, DEPARTMENT),
], nvars=3, required_vars=[X, Y, Z])

# Q2 — 1< 2>
Q2 = Plan([
    (COURSE, TYPE_ID, DIR_IN, X),
    (X, NAME, DIR_OUT, Y),
], nvars=2, required_vars=[X, Y])

# Q3 — 1<< 2> 3> 4> 5> 6> 1>   (result is empty by schema: UG students
# have no undergraduateDegreeFrom — matches the reference's #R=0)
Q3 = Plan([
    (UGDEGREE, PREDICATE_ID, DIR_IN, X),    # predicate-index start
    (X, TYPE_ID, DIR_OUT, UGSTUDENT),
    (X, MEMBEROF, DIR_OUT, Z),
    (Z, TYPE_ID, DIR_OUT, DEPARTMENT),
    (Z, SUBORG, DIR_OUT, Y),
    (Y, TYPE_ID, DIR_OUT, UNIVERSITY),
    (X, UGDEGREE, DIR_OUT, Y),              # k2k
], nvars=3, required_vars=[X, Y, Z])

# Q4 — 1< 2> 3> 4> 5>
Q4 = Plan([
    (DEPT0_UNIV0, WORKSFOR, DIR_IN, X),     # const start
    (X, TYPE_ID, DIR_OUT, FULLPROF),
    (X, NAME, DIR_OUT, Y1),
    (X, EMAIL, DIR_OUT, Y2),
    (X, TELEPHONE, DIR_OUT, Y3),
], nvars=4, required_vars=[X, Y1, Y2, Y3])

# Q5 — 1< 2>
Q5 = Plan([
    (DEPT0_UNIV0, SUBORG, DIR_IN, X),
    (X, TYPE_ID, DIR_OUT, RESEARCHGROUP),
], nvars=1, required_vars=[X])

# Q6 — 1< 2> 3< 4>
Q6 = Plan([
    (UNIV0, SUBORG, DIR_IN, Y),
    (Y, TYPE_ID, DIR_OUT, DEPARTMENT),
    (Y, WORKSFOR, DIR_IN, X),
    (X, TYPE_ID, DIR_OUT, FULLPROF),
], nvars=2, required_vars=[X, Y])

# Q7 — 1< 2< 3> 4> 5> 6<
Q7 = Plan([
    (FULLPROF, TYPE_ID, DIR_IN, Y),
    (Y, ADVISOR, DIR_IN, X),
    (X, TYPE_ID, DIR_OUT, UGSTUDENT),
    (X, TAKESCOURSE, DIR_OUT, Z),
    (Z, TYPE_ID, DIR_OUT, COURSE),
    (Z, TEACHEROF, DIR_IN, Y),              # k2k
], nvars=3, required_vars=[X, Y, Z])

ALL = {"q1": Q1, "q2": Q2, "q3": Q3, "q4": Q4, "q5": Q5, "q6": Q6, "q7": Q7}
HEAVY = ["q1", "q2", "q3", "q7"]
LIGHT = ["q4", "q5", "q6"]


def emu_template(name, const):
    """Light emulator templates A1-A6 instantiated with a candidate
    constant (scripts/sparql_query/lubm/emulator/q{1..6}; %-placeholders
    filled from type-index candidates like Proxy::fill_template,
    core/proxy.hpp:69-129)."""
    t = {
        "a1": Plan([(const, TAKESCOURSE, DIR_IN, X),
                    (X, TYPE_ID, DIR_OUT, GRADSTUDENT)], 1, [X]),
        "a2": Plan([(const, PUBAUTHOR, DIR_IN, X),
                    (X, TYPE_ID, DIR_OUT, PUBLICATION)], 1, [X]),
        "a3": Plan([(const, MEMBEROF, DIR_IN, X),
                    (X, TYPE_ID, DIR_OUT, GRADSTUDENT)], 1, [X]),
        "a4": Plan([(const, WORKSFOR, DIR_IN, X),
                    (X, TYPE_ID, DIR_OUT, FULLPROF),
                    (X, NAME, DIR_OUT, Y1),
                    (X, EMAIL, DIR_OUT, Y2),
                    (X, TELEPHONE, DIR_OUT, Y3)], 4, [X, Y1, Y2, Y3]),
        "a5": Plan([(const, SUBORG, DIR_IN, X),
                    (X, TYPE_ID, DIR_OUT, RESEARCHGROUP)], 1, [X]),
        "a6": Plan([(const, SUBORG, DIR_IN, Y),
                    (Y, TYPE_ID, DIR_OUT, DEPARTMENT),
                    (Y, WORKSFOR, DIR_IN, X),
                    (X, TYPE_ID, DIR_OUT, FULLPROF)], 2, [X, Y]),
    }
    return t[name]


# candidate pool type per template + mix weights
# (scripts/sparql_query/lubm/emulator/mix_config: 25/25/3/6/25/2)
EMU_POOLS = {"a1": GRADCOURSE, "a2": ASSTPROF, "a3": DEPARTMENT,
             "a4": DEPARTMENT, "a5": DEPARTMENT, "a6": UNIVERSITY}
EMU_WEIGHTS = {"a1": 25, "a2": 25, "a3": 3, "a4": 6, "a5": 25, "a6": 2}


def versatile_plans(store):
    """LUBM Q8-Q12 — the VERSATILE (predicate-variable) queries
    (scripts/sparql_query/lubm/basic/lubm_q{8..12}; operators
    sparql.hpp:556-744).  The reference names its constants by URI
    (Department3.University0/FullProfessor1, ...); our generator ids are
    allocation-order, so the constants are derived positionally from the
    store: dept k of university 0, the first FullProfessor of dept 3, an
    UndergraduateStudent member of dept 7.  Same shapes/selectivity."""
    import numpy as np

    def tris(v, p, d):
        return np.sort(np.asarray(store.get_triples(int(v), p, d)))

    def first_of_type(cands, t):
        for c in cands:
            if t in tris(c, TYPE_ID, DIR_OUT):
                return int(c)
        raise RuntimeError("no candidate of requested type")

    depts = tris(UNIV0, SUBORG, DIR_IN)
    dept3 = int(depts[3 % len(depts)])
    dept7 = int(depts[7 % len(depts)])
    prof = first_of_type(tris(dept3, WORKSFOR, DIR_IN), FULLPROF)
    members = tris(dept7, MEMBEROF, DIR_IN)
    ug = [int(c) for c in members if UGSTUDENT in tris(c, TYPE_ID, DIR_OUT)]
    student = ug[201 % len(ug)]

    # Q8: ?D subOrgOf Univ0 . ?X ?P ?D  (known_unknown_unknown, d=IN)
    q8 = Plan([(UNIV0, SUBORG, DIR_IN, -1),
               (-1, -2, DIR_IN, -3)], 3, [-3, -2, -1])
    # Q9: ?X ?P Dept0.Univ0  (const_unknown_unknown, d=IN)
    q9 = Plan([(DEPT0_UNIV0, -1, DIR_IN, -2)], 2, [-2, -1])
    # Q10: ?X type UGStudent . ?X ?Y Prof  (planner starts from the
    # constant: const_unknown_unknown + typeof filter)
    q10 = Plan([(prof, -1, DIR_IN, -2),
                (-2, TYPE_ID, DIR_OUT, UGSTUDENT)], 2, [-2, -1])
    # Q11: Student ?X Dept7  (const_unknown_const)
    q11 = Plan([(student, -1, DIR_OUT, dept7)], 1, [-1])
    # Q12: ?Y subOrgOf Univ0 . ?X worksFor ?Y . ?Z advisor ?X
    # (fixed predicates; ?Z is generated but not selected)
    q12 = Plan([(UNIV0, SUBORG, DIR_IN, -1),
                (-1, WORKSFOR, DIR_IN, -2),
                (-2, ADVISOR, DIR_IN, -3)], 3, [-2, -1])
    return {"q8": q8, "q9": q9, "q10": q10, "q11": q11, "q12": q12}
