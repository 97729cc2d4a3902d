"""Minimal SPARQL parser + plan application (control plane, above the
hot path).

Restates the reference's behavior for the benchmark query class
(core/parser.hpp + core/SPARQLParser.hpp basic-graph-pattern subset):
PREFIX declarations, SELECT var list, WHERE { s p o . ... } triples —
plus the planner-plan format of scripts/sparql_query/*/osdi16_plan/*.fmt
(documented in those files' headers):

    `N >`  — pattern N as written: (s, p, OUT, o)
    `N <`  — pattern N reversed:   (o, p, IN, s)
    `N <<` — start from the predicate index: (p, PREDICATE_ID, IN, s)
    `N >>` — predicate index, object side:   (p, PREDICATE_ID, OUT, o)

Strings map to ids through a vocabulary dict (the synthetic datasets
assign ids deterministically; the reference's StringServer str_normal /
str_index files play this role, core/string_server.hpp — out of scope
per SURVEY.md §2)."""
import re

from . import Plan

PREDICATE_ID, TYPE_ID = 0, 1
DIR_IN, DIR_OUT = 0, 1


def lubm_vocab():
    """Vocabulary of the LUBM-shaped generator (lubm_gen.cpp ids)."""
    from . import queries as Q
    v = {
        "rdf:type": TYPE_ID,
        "ub:subOrganizationOf": Q.SUBORG,
        "ub:undergraduateDegreeFrom": Q.UGDEGREE,
        "ub:memberOf": Q.MEMBEROF,
        "ub:worksFor": Q.WORKSFOR,
        "ub:teacherOf": Q.TEACHEROF,
        "ub:advisor": Q.ADVISOR,
        "ub:takesCourse": Q.TAKESCOURSE,
        "ub:name": Q.NAME,
        "ub:emailAddress": Q.EMAIL,
        "ub:telephone": Q.TELEPHONE,
        "ub:headOf": Q.HEADOF,
        "ub:doctoralDegreeFrom": Q.DOCDEGREE,
        "ub:publicationAuthor": Q.PUBAUTHOR,
        "ub:University": Q.UNIVERSITY,
        "ub:Department": Q.DEPARTMENT,
        "ub:FullProfessor": Q.FULLPROF,
        "ub:AssociateProfessor": Q.ASSOCPROF,
        "ub:AssistantProfessor": Q.ASSTPROF,
        "ub:Lecturer": Q.LECTURER,
        "ub:UndergraduateStudent": Q.UGSTUDENT,
        "ub:GraduateStudent": Q.GRADSTUDENT,
        "ub:Course": Q.COURSE,
        "ub:GraduateCourse": Q.GRADCOURSE,
        "ub:ResearchGroup": Q.RESEARCHGROUP,
        "ub:Publication": Q.PUBLICATION,
        # well-known entity IRIs of the synthetic dataset
        "<http://www.University0.edu>": Q.UNIV0,
        "<http://www.Department0.University0.edu>": Q.DEPT0_UNIV0,
    }
    return v


class ParseError(ValueError):
    pass


def parse(text, vocab, plan_lines=None, distinct=None, limit=None,
          offset=None):
    """SPARQL text -> Plan.  plan_lines: optional .fmt directives
    (list of strings like '3 <'), mirroring Planner::set_plan."""
    # strip comments
    text = re.sub(r"#[^\n]*", "", text)
    prefixes = dict(re.findall(r"PREFIX\s+(\w+:)\s*<([^>]*)>", text))
    m = re.search(r"SELECT\s+(DISTINCT\s+)?(.*?)\s+WHERE\s*\{(.*)\}",
                  text, re.S | re.I)
    if not m:
        raise ParseError("no SELECT ... WHERE { ... } found")
    if distinct is None:
        distinct = bool(m.group(1))
    sel = m.group(2).split()
    body = m.group(3)

    # OPTIONAL { ... } group and { A } UNION { B } branches (SPARQL
    # subset matching the engine's support: one OPTIONAL BGP, one
    # UNION chain — query.hpp:708-733)
    opt_body = None
    union_bodies = []
    om2 = re.search(r"OPTIONAL\s*\{([^{}]*)\}", body, re.I | re.S)
    if om2:
        opt_body = om2.group(1)
        body = body[:om2.start()] + body[om2.end():]
    um = re.search(
        r"\{([^{}]*)\}(\s*UNION\s*\{([^{}]*)\})+", body, re.I | re.S)
    if um:
        whole = um.group(0)
        union_bodies = re.findall(r"\{([^{}]*)\}", whole)
        body = body[:um.start()] + body[um.end():]
    lm = re.search(r"LIMIT\s+(\d+)", text, re.I)
    om = re.search(r"OFFSET\s+(\d+)", text, re.I)
    if limit is None:
        limit = int(lm.group(1)) if lm else -1
    if offset is None:
        offset = int(om.group(1)) if om else 0

    vars_ = {}

    def var_id(tok):
        if tok not in vars_:
            vars_[tok] = -(len(vars_) + 1)
        return vars_[tok]

    def term(tok):
        if tok.startswith("?"):
            return var_id(tok)
        if tok in vocab:
            return vocab[tok]
        resolver = vocab.get("__resolver__")
        if resolver is not None:
            v = resolver(tok)
            if v is not None:
                return v
        if ":" in tok and tok.split(":")[0] + ":" in prefixes:
            raise ParseError(f"unknown vocabulary term {tok!r}")
        if tok.startswith("<") and tok.endswith(">"):
            raise ParseError(f"unknown IRI {tok!r}")
        raise ParseError(f"cannot resolve term {tok!r}")

    def split_triples(text_body):
        out = []
        cur2 = []
        for tok in text_body.split():
            # '.' separates patterns (IRIs contain dots, so split on the
            # standalone token, not on the character)
            if tok == ".":
                if cur2:
                    if len(cur2) != 3:
                        raise ParseError(
                            f"expected triple pattern, got {cur2!r}")
                    out.append(tuple(cur2))
                    cur2 = []
                continue
            if tok.endswith(".") and not tok.endswith(">."):
                cur2.append(tok[:-1])
                if len(cur2) != 3:
                    raise ParseError(f"expected triple pattern, got {cur2!r}")
                out.append(tuple(cur2))
                cur2 = []
                continue
            cur2.append(tok)
        if cur2:
            if len(cur2) != 3:
                raise ParseError(f"expected triple pattern, got {cur2!r}")
            out.append(tuple(cur2))
        return out

    raw = split_triples(body)

    def resolve(i, direction):
        s, p, o = raw[i]
        if direction == ">":
            return (term(s), term(p), DIR_OUT, term(o))
        if direction == "<":
            return (term(o), term(p), DIR_IN, term(s))
        if direction == "<<":  # predicate-index start on the subject side
            return (term(p), PREDICATE_ID, DIR_IN, term(s))
        if direction == ">>":  # predicate-index start on the object side
            return (term(p), PREDICATE_ID, DIR_OUT, term(o))
        raise ParseError(f"bad plan direction {direction!r}")

    patterns = []
    if plan_lines:
        for line in plan_lines:
            line = line.split("#")[0].strip()
            if not line:
                continue
            n, d = line.split()
            patterns.append(resolve(int(n) - 1, d))
    else:
        # unplanned: textual order, forward direction (the reference
        # requires a planner or a .fmt for good plans; so do we)
        for i in range(len(raw)):
            patterns.append(resolve(i, ">"))

    # rdf:type with a constant object, reversed, is a TYPE-index start:
    # the .fmt 'N <' on "?X rdf:type T" yields (T, TYPE_ID, IN, ?X) which
    # start_from_index() recognises (query.hpp:660-682) — nothing special
    # to do here, the ids compose.
    def to_pats(text_body):
        return [(term(a), term(b), DIR_OUT, term(c))
                for (a, b, c) in split_triples(text_body)]

    optional = to_pats(opt_body) if opt_body else []
    unions = [to_pats(b) for b in union_bodies] if union_bodies else []

    req = [vars_[v] for v in sel]
    return Plan(patterns, nvars=len(vars_), required_vars=req,
                distinct=distinct, limit=limit, offset=offset,
                optional=optional, unions=unions)


def lubm_entity_vocab(store):
    """lubm_vocab + a positional entity-URI resolver: the reference maps
    URIs through its StringServer; our synthetic ids are allocation-
    order, so DepartmentK.UniversityN / FullProfessorJ / ... resolve
    positionally through the store (same shape and selectivity)."""
    from . import queries as Q
    import numpy as np

    def tris(v, p, d):
        return np.sort(np.asarray(store.get_triples(int(v), p, d)))

    cache = {}

    def depts(n):
        if ("d", n) not in cache:
            cache[("d", n)] = tris((n + 1) << 17, Q.SUBORG, DIR_IN)
        return cache[("d", n)]

    def members_of_type(dept, pred, t):
        key = ("m", dept, pred, t)
        if key not in cache:
            cache[key] = [int(c) for c in tris(dept, pred, DIR_IN)
                          if t in tris(c, TYPE_ID, DIR_OUT)]
        return cache[key]

    import re as _re

    def resolver(tok):
        m = _re.fullmatch(r"<http://www\.University(\d+)\.edu>", tok)
        if m:
            return (int(m.group(1)) + 1) << 17
        m = _re.fullmatch(
            r"<http://www\.Department(\d+)\.University(\d+)\.edu>", tok)
        if m:
            ds = depts(int(m.group(2)))
            return int(ds[int(m.group(1)) % len(ds)])
        m = _re.fullmatch(
            r"<http://www\.Department(\d+)\.University(\d+)\.edu/"
            r"(FullProfessor|UndergraduateStudent|GraduateStudent)(\d+)>",
            tok)
        if m:
            ds = depts(int(m.group(2)))
            dept = int(ds[int(m.group(1)) % len(ds)])
            kind, j = m.group(3), int(m.group(4))
            if kind == "FullProfessor":
                pool = members_of_type(dept, Q.WORKSFOR, Q.FULLPROF)
            elif kind == "UndergraduateStudent":
                pool = members_of_type(dept, Q.MEMBEROF, Q.UGSTUDENT)
            else:
                pool = members_of_type(dept, Q.MEMBEROF, Q.GRADSTUDENT)
            return pool[j % len(pool)] if pool else None
        return None

    v = lubm_vocab()
    v["__resolver__"] = resolver
    return v
