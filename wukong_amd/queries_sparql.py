"""LUBM Q1-Q7 as SPARQL text + the OSDI16 plan orders, for the text
parser (wukong_amd.sparql).  Queries are the standard LUBM benchmark
patterns over the univ-bench ontology; plan orders restate
scripts/sparql_query/lubm/basic/osdi16_plan/*.fmt."""

_P = ("PREFIX rdf: <http://www.w3.org/1999/02/22-rdf-syntax-ns#>\n"
      "PREFIX ub: <http://swat.cse.lehigh.edu/onto/univ-bench.owl#>\n")

TEXT = {
    "q1": _P + """SELECT ?X ?Y ?Z WHERE {
        ?Y rdf:type ub:University .
        ?X ub:undergraduateDegreeFrom ?Y .
        ?X rdf:type ub:GraduateStudent .
        ?X ub:memberOf ?Z .
        ?Z ub:subOrganizationOf ?Y .
        ?Z rdf:type ub:Department .
    }""",
    "q2": _P + """SELECT ?X ?Y WHERE {
        ?X rdf:type ub:Course .
        ?X ub:name ?Y .
    }""",
    "q3": _P + """SELECT ?X ?Y ?Z WHERE {
        ?X ub:undergraduateDegreeFrom ?Y .
        ?X rdf:type ub:UndergraduateStudent .
        ?X ub:memberOf ?Z .
        ?Z rdf:type ub:Department .
        ?Z ub:subOrganizationOf ?Y .
        ?Y rdf:type ub:University .
    }""",
    "q4": _P + """SELECT ?X ?Y1 ?Y2 ?Y3 WHERE {
        ?X ub:worksFor <http://www.Department0.University0.edu> .
        ?X rdf:type ub:FullProfessor .
        ?X ub:name ?Y1 .
        ?X ub:emailAddress ?Y2 .
        ?X ub:telephone ?Y3 .
    }""",
    "q5": _P + """SELECT ?X WHERE {
        ?X ub:subOrganizationOf <http://www.Department0.University0.edu> .
        ?X rdf:type ub:ResearchGroup .
    }""",
    "q6": _P + """SELECT ?X ?Y WHERE {
        ?Y ub:subOrganizationOf <http://www.University0.edu> .
        ?Y rdf:type ub:Department .
        ?X ub:worksFor ?Y .
        ?X rdf:type ub:FullProfessor .
    }""",
    "q7": _P + """SELECT ?X ?Y ?Z WHERE {
        ?Y rdf:type ub:FullProfessor .
        ?X ub:advisor ?Y .
        ?X rdf:type ub:UndergraduateStudent .
        ?X ub:takesCourse ?Z .
        ?Z rdf:type ub:Course .
        ?Y ub:teacherOf ?Z .
    }""",
}

# osdi16_plan/*.fmt orders (1-based pattern indices + direction)
PLAN = {
    "q1": ["3 <", "4 >", "2 >", "1 >", "5 <", "6 >"],
    "q2": ["1 <", "2 >"],
    "q3": ["1 <<", "2 >", "3 >", "4 >", "5 >", "6 >", "1 >"],
    "q4": ["1 <", "2 >", "3 >", "4 >", "5 >"],
    "q5": ["1 <", "2 >"],
    "q6": ["1 <", "2 >", "3 <", "4 >"],
    "q7": ["1 <", "2 <", "3 >", "4 >", "5 >", "6 <"],
}


# Q8-Q12 — the VERSATILE (predicate-variable) queries
# (scripts/sparql_query/lubm/basic/lubm_q{8..12}).  No OSDI16 plans
# exist for these; they go through the greedy planner
# (wukong_amd.planner).  Entity URIs resolve positionally through
# sparql.lubm_entity_vocab(store).
TEXT_VERSATILE = {
    "q8": _P + """SELECT ?X ?P ?D WHERE {
        ?D ub:subOrganizationOf <http://www.University0.edu> .
        ?X ?P ?D .
    }""",
    "q9": _P + """SELECT ?X ?P WHERE {
        ?X ?P <http://www.Department0.University0.edu> .
    }""",
    "q10": _P + """SELECT ?X ?Y WHERE {
        ?X rdf:type ub:UndergraduateStudent .
        ?X ?Y <http://www.Department3.University0.edu/FullProfessor1> .
    }""",
    "q11": _P + """SELECT ?X WHERE {
        <http://www.Department7.University0.edu/UndergraduateStudent201> ?X <http://www.Department7.University0.edu> .
    }""",
    "q12": _P + """SELECT ?X ?Y WHERE {
        ?Y ub:subOrganizationOf <http://www.University0.edu> .
        ?X ub:worksFor ?Y .
        ?Z ub:advisor ?X .
    }""",
}
