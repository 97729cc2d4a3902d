/*
 * lubm_gen.cpp — seeded LUBM-shaped synthetic ID-triple generator.
 *
 * Replaces the absent Java LUBM generator + datagen/generate_data.cpp
 * (NT→ID conversion).  ID scheme follows datagen/generate_data.cpp:122-123:
 * index ids (predicates AND types) from 2, normal vertex ids from 2^17.
 * Schema and cardinalities follow the LUBM ontology as used by the
 * reference's Q1-Q7 (scripts/sparql_query/lubm/basic): notably
 * UndergraduateStudents have NO undergraduateDegreeFrom (so Q3 yields 0
 * rows, matching docs/performance/S1C24-LUBM2560-20181203.md:125), and
 * GraduateStudents draw undergraduateDegreeFrom uniformly over all
 * universities (Q1 selectivity ~1/U, matching the reference's 2528 rows
 * at LUBM-2560 ~= 1/univ).
 *
 * Deterministic: university u is generated from splitmix64(seed, u)
 * regardless of partitioning, so every (sid, nsrv) sees the same graph.
 * University u's entities live in id block [(u+1)<<17, (u+2)<<17).
 */
#include "wk_types.h"
#include <vector>
#include <cstdlib>
#include <cstring>
#include <cstdio>
#include <omp.h>

using namespace wk;

namespace lubm {

// ---- fixed schema enumeration (stable ids; documented in DESIGN.md) ----
enum : sid_t {
    P_SUBORG = 2,      // ub:subOrganizationOf
    P_UGDEGREE = 3,    // ub:undergraduateDegreeFrom
    P_MEMBEROF = 4,    // ub:memberOf
    P_WORKSFOR = 5,    // ub:worksFor
    P_TEACHEROF = 6,   // ub:teacherOf
    P_ADVISOR = 7,     // ub:advisor
    P_TAKESCOURSE = 8, // ub:takesCourse
    P_NAME = 9,        // ub:name
    P_EMAIL = 10,      // ub:emailAddress
    P_TELEPHONE = 11,  // ub:telephone
    P_HEADOF = 12,     // ub:headOf
    P_DOCDEGREE = 13,  // ub:doctoralDegreeFrom
    T_UNIVERSITY = 14,
    T_DEPARTMENT = 15,
    T_FULLPROF = 16,
    T_ASSOCPROF = 17,
    T_ASSTPROF = 18,
    T_LECTURER = 19,
    T_UGSTUDENT = 20,
    T_GRADSTUDENT = 21,
    T_COURSE = 22,
    T_GRADCOURSE = 23,
    T_RESEARCHGROUP = 24,
    P_PUBAUTHOR = 25,  // ub:publicationAuthor (emulator template A2)
    T_PUBLICATION = 26,
    P_MASTERSDEGREE = 27,  // ub:mastersDegreeFrom
    P_RESEARCHINT = 28,    // ub:researchInterest
};

struct rng_t {  // splitmix64
    uint64_t s;
    uint64_t next() {
        uint64_t z = (s += 0x9e3779b97f4a7c15ull);
        z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
        z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
        return z ^ (z >> 31);
    }
    // uniform in [lo, hi)
    uint32_t range(uint32_t lo, uint32_t hi) { return lo + (uint32_t)(next() % (hi - lo)); }
};

struct emitter {
    std::vector<sid_t> &out;
    int sid, nsrv;
    void emit(sid_t s, sid_t p, sid_t o) {
        // partition filter mirrors core/loader/base_loader.hpp:344-352
        if ((int)(s % (sid_t)nsrv) == sid || (int)(o % (sid_t)nsrv) == sid) {
            out.push_back(s); out.push_back(p); out.push_back(o);
        }
    }
};

static void gen_university(uint32_t u, uint32_t nuniv, uint64_t seed, emitter &em) {
    rng_t rng{hash_u64(seed * 0x9e3779b97f4a7c15ull + u + 1)};
    uint64_t base = ((uint64_t)u + 1) << NBITS_IDX;
    uint64_t next_id = base;
    auto alloc = [&]() { return (sid_t)next_id++; };

    sid_t univ = alloc();
    em.emit(univ, TYPE_ID, T_UNIVERSITY);
    em.emit(univ, P_NAME, alloc());

    auto rand_univ_id = [&]() {  // some OTHER (or same) university's entity id
        uint32_t v = rng.range(0, nuniv);
        return (sid_t)(((uint64_t)v + 1) << NBITS_IDX);  // the university entity itself
    };

    int ndept = rng.range(15, 26);
    for (int d = 0; d < ndept; d++) {
        sid_t dept = alloc();
        em.emit(dept, TYPE_ID, T_DEPARTMENT);
        em.emit(dept, P_SUBORG, univ);
        em.emit(dept, P_NAME, alloc());

        // faculty (real-LUBM degree shape: ug + masters + doctoral for
        // professors, ug + masters for lecturers; researchInterest all)
        std::vector<sid_t> profs, faculty;
        int nfull = rng.range(7, 11), nassoc = rng.range(10, 15),
            nasst = rng.range(8, 12), nlect = rng.range(5, 8);
        auto person = [&](sid_t type, bool is_prof) {
            sid_t p = alloc();
            em.emit(p, TYPE_ID, type);
            em.emit(p, P_WORKSFOR, dept);
            em.emit(p, P_NAME, alloc());
            em.emit(p, P_EMAIL, alloc());
            em.emit(p, P_TELEPHONE, alloc());
            em.emit(p, P_UGDEGREE, rand_univ_id());
            em.emit(p, P_MASTERSDEGREE, rand_univ_id());
            em.emit(p, P_RESEARCHINT, alloc());
            if (is_prof) {
                em.emit(p, P_DOCDEGREE, rand_univ_id());
                profs.push_back(p);
            }
            faculty.push_back(p);
            return p;
        };
        for (int i = 0; i < nfull; i++) person(T_FULLPROF, true);
        for (int i = 0; i < nassoc; i++) person(T_ASSOCPROF, true);
        for (int i = 0; i < nasst; i++) person(T_ASSTPROF, true);
        for (int i = 0; i < nlect; i++) person(T_LECTURER, false);
        em.emit(profs[0], P_HEADOF, dept);

        // courses: each faculty member teaches 1-2 undergraduate AND 1-2
        // graduate courses (undergraduate Course count per dept ~54
        // matches the published LUBM-2560 Q2 #R = 2,765,067 — one Course
        // row per course, S1C24-LUBM2560-20181203.md:124)
        std::vector<sid_t> courses, gradcourses;
        for (sid_t f : faculty) {
            int nug_c = rng.range(1, 3), ngrad_c = rng.range(1, 3);
            for (int i = 0; i < nug_c + ngrad_c; i++) {
                sid_t c = alloc();
                bool grad = i >= nug_c;
                em.emit(c, TYPE_ID, grad ? T_GRADCOURSE : T_COURSE);
                em.emit(c, P_NAME, alloc());
                em.emit(f, P_TEACHEROF, c);
                (grad ? gradcourses : courses).push_back(c);
            }
        }
        if (courses.empty()) {  // ensure both pools non-empty
            sid_t c = alloc();
            em.emit(c, TYPE_ID, T_COURSE); em.emit(c, P_NAME, alloc());
            em.emit(faculty[0], P_TEACHEROF, c); courses.push_back(c);
        }
        if (gradcourses.empty()) {
            sid_t c = alloc();
            em.emit(c, TYPE_ID, T_GRADCOURSE); em.emit(c, P_NAME, alloc());
            em.emit(faculty[0], P_TEACHEROF, c); gradcourses.push_back(c);
        }

        // undergraduate students (NO ugDegreeFrom: Q3 must yield 0 rows);
        // counts + 2/9 advisor fraction calibrated so Q7 (UG taking a
        // course taught by their advisor) lands near the published
        // 112,559 at LUBM-2560
        int nug = (int)faculty.size() * rng.range(9, 16);
        for (int i = 0; i < nug; i++) {
            sid_t s = alloc();
            em.emit(s, TYPE_ID, T_UGSTUDENT);
            em.emit(s, P_MEMBEROF, dept);
            em.emit(s, P_NAME, alloc());
            em.emit(s, P_EMAIL, alloc());
            em.emit(s, P_TELEPHONE, alloc());
            int nc = rng.range(2, 5);
            for (int k = 0; k < nc; k++)
                em.emit(s, P_TAKESCOURSE, courses[rng.range(0, courses.size())]);
            if (rng.next() % 9u < 2)  // 2/9 of UG students have an advisor
                em.emit(s, P_ADVISOR, profs[rng.range(0, profs.size())]);
        }

        // graduate students
        int ngrad = (int)faculty.size() * rng.range(3, 5);
        for (int i = 0; i < ngrad; i++) {
            sid_t s = alloc();
            em.emit(s, TYPE_ID, T_GRADSTUDENT);
            em.emit(s, P_MEMBEROF, dept);
            em.emit(s, P_NAME, alloc());
            em.emit(s, P_EMAIL, alloc());
            em.emit(s, P_TELEPHONE, alloc());
            em.emit(s, P_UGDEGREE, rand_univ_id());
            int nc = rng.range(1, 4);
            for (int k = 0; k < nc; k++)
                em.emit(s, P_TAKESCOURSE, gradcourses[rng.range(0, gradcourses.size())]);
            em.emit(s, P_ADVISOR, profs[rng.range(0, profs.size())]);
        }

        // publications (real-LUBM per-rank ranges: FullProf 15-20,
        // AssocProf 10-17, AsstProf 5-10, Lecturer 0-5), each with a
        // name literal; half with one professor co-author (emulator A2
        // shape: %AssistantProfessor pubs)
        {
            int fi = 0;
            auto pubs_for = [&](sid_t f, int lo, int hi) {
                int np = rng.range(lo, hi);
                for (int i = 0; i < np; i++) {
                    sid_t pub = alloc();
                    em.emit(pub, TYPE_ID, T_PUBLICATION);
                    em.emit(pub, P_NAME, alloc());
                    em.emit(pub, P_PUBAUTHOR, f);
                    if (rng.next() & 1)
                        em.emit(pub, P_PUBAUTHOR, profs[rng.range(0, profs.size())]);
                }
            };
            for (int i = 0; i < nfull; i++) pubs_for(faculty[fi++], 15, 21);
            for (int i = 0; i < nassoc; i++) pubs_for(faculty[fi++], 10, 18);
            for (int i = 0; i < nasst; i++) pubs_for(faculty[fi++], 5, 11);
            for (int i = 0; i < nlect; i++) pubs_for(faculty[fi++], 0, 6);
        }

        // research groups
        int nrg = rng.range(10, 21);
        for (int i = 0; i < nrg; i++) {
            sid_t g = alloc();
            em.emit(g, TYPE_ID, T_RESEARCHGROUP);
            em.emit(g, P_SUBORG, dept);
        }
    }

    if (next_id - base >= (1ull << NBITS_IDX)) {
        fprintf(stderr, "lubm_gen: university %u overflowed its id block (%lu)\n",
                u, (unsigned long)(next_id - base));
        abort();
    }
}

}  // namespace lubm

extern "C" int64_t wk_lubm_gen(int32_t nuniv, uint64_t seed, int32_t sid,
                               int32_t nsrv, sid_t **out_spo) {
    if (nuniv <= 0 || nsrv <= 0 || sid < 0 || sid >= nsrv || !out_spo) return -1;
    int nthr = omp_get_max_threads();
    std::vector<std::vector<sid_t>> parts(nuniv);
#pragma omp parallel for schedule(dynamic, 4) num_threads(nthr)
    for (int u = 0; u < nuniv; u++) {
        parts[u].reserve(137000 * 3 / nsrv + 1024);
        lubm::emitter em{parts[u], sid, nsrv};
        lubm::gen_university((uint32_t)u, (uint32_t)nuniv, seed, em);
    }
    int64_t total = 0;
    std::vector<int64_t> offs(nuniv + 1, 0);
    for (int u = 0; u < nuniv; u++) { offs[u] = total; total += (int64_t)parts[u].size(); }
    offs[nuniv] = total;
    sid_t *buf = (sid_t *)malloc((size_t)total * sizeof(sid_t));
    if (!buf) return -1;
#pragma omp parallel for schedule(dynamic, 4)
    for (int u = 0; u < nuniv; u++)
        memcpy(buf + offs[u], parts[u].data(), parts[u].size() * sizeof(sid_t));
    *out_spo = buf;
    return total / 3;
}

extern "C" void wk_free_triples(sid_t *spo) { free(spo); }
