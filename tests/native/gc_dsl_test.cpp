// The reference's own GarbageSkipper golden vectors transcribed verbatim:
// CompactionIteratorTest.java:93-176 (testGcCompactionSupersede{Left,Middle,
// Right}, SwitchInSuperseded, Boundaries, Matches, RowDeletion,
// PartitionDeletion), using UnfilteredRowsGenerator.parse's DSL
// (UnfilteredRowsGenerator.java:187-221): "N<[T]"/"N<=[T]" open marker,
// "[T]<N"/"[T]<=N" close marker, "N[TS]"/"N[TSDdel]" no-cell row, "Dxx|"
// partition deletion; adjacent close+open at one position become boundary
// markers (attachBoundaries). Harness constants: NOW=1000, GC_BEFORE=100.
// Pipeline under test: oracle merge -> garbage_filter (ROW mode) -> purge,
// checked for (a) the reference's expected output / size bound and (b) the
// reference's equivalence law: merge(inputs+tombs) == merge(result+tombs).
#include "../../oracle/src/compact.h"
#include <cstdio>
#include <cstring>
#include <regex>
#include <sstream>
#include <string>
#include <vector>

using namespace oracle;

static int fails = 0;
#define CHECK(cond, msg)                                            \
    do {                                                            \
        if (!(cond)) { printf("FAIL: %s\n", msg); fails++; }        \
        else printf("ok:   %s\n", msg);                             \
    } while (0)

static bytes be8(int64_t v) {
    bytes b(8);
    for (int i = 0; i < 8; i++) b[i] = (uint8_t)((uint64_t)v >> (8 * (7 - i)));
    return b;
}

static Header dsl_header() {
    Header h;
    h.key_type = CqlType::LONG;
    h.clustering_types = {CqlType::LONG};
    h.regular_cols = {{bytes{'v'}, CqlType::BYTES}};
    h.stats.min_ts = 100;
    h.stats.min_ldt = 100;
    h.stats.min_ttl = 0;
    return h;
}

static Partition parse_dsl(std::string in) {
    Partition p;
    p.key = bytes{'k'};
    p.set_token();
    std::smatch m;
    if (std::regex_search(in, m, std::regex(R"(^D(\d+)\|)"))) {
        int64_t d = std::stoll(m[1]);
        p.del = DeletionTime{d, (uint32_t)d};
        in = in.substr(m[0].length());
    }
    std::regex open(R"((\d+)<(=)?\[(\d+)\])"), close(R"(\[(\d+)\]<(=)?(\d+))"),
        row(R"((\d+)(\[(\d+)(?:D(\d+))?\])?)");
    std::istringstream ss(in);
    std::string tok;
    while (ss >> tok) {
        if (std::regex_match(tok, m, open)) {
            Unfiltered u;
            u.kind = Unfiltered::MARKER;
            u.marker.kind = m[2].matched ? INCL_START : EXCL_START;
            u.marker.values = {ClusterVal{ClusterVal::VALUE, be8(std::stoll(m[1]))}};
            int64_t t = std::stoll(m[3]);
            u.marker.end_dt = DeletionTime{t, (uint32_t)t};
            p.items.push_back(std::move(u));
        } else if (std::regex_match(tok, m, close)) {
            Unfiltered u;
            u.kind = Unfiltered::MARKER;
            u.marker.kind = m[2].matched ? INCL_END : EXCL_END;
            u.marker.values = {ClusterVal{ClusterVal::VALUE, be8(std::stoll(m[3]))}};
            int64_t t = std::stoll(m[1]);
            u.marker.end_dt = DeletionTime{t, (uint32_t)t};
            p.items.push_back(std::move(u));
        } else if (std::regex_match(tok, m, row)) {
            Row r;
            r.clustering = {ClusterVal{ClusterVal::VALUE, be8(std::stoll(m[1]))}};
            r.live.ts = m[3].matched ? std::stoll(m[3]) : 999;  // default NOW-1
            if (m[4].matched) {
                int64_t d = std::stoll(m[4]);
                r.del = DeletionTime{d, (uint32_t)d};
            }
            r.cells.resize(1);
            Unfiltered u;
            u.kind = Unfiltered::ROW;
            u.row = std::move(r);
            p.items.push_back(std::move(u));
        } else {
            printf("FAIL: can't parse DSL token '%s'\n", tok.c_str());
            fails++;
        }
    }
    // attachBoundaries: close marker + open marker at the same position fold
    // into a boundary (UnfilteredRowsGenerator.attachBoundaries semantics:
    // the pair must form a continuous bound — INCL_END+EXCL_START or
    // EXCL_END+INCL_START at equal clustering values)
    std::vector<Unfiltered> out;
    for (auto& u : p.items) {
        if (!out.empty() && out.back().kind == Unfiltered::MARKER &&
            u.kind == Unfiltered::MARKER && !out.back().marker.boundary()) {
            Marker& a = out.back().marker;
            const Marker& b = u.marker;
            bool same_pos = a.values[0].v == b.values[0].v;
            if (same_pos && a.kind == INCL_END && b.kind == EXCL_START) {
                a.kind = INCL_END_EXCL_START;
                a.start_dt = b.end_dt;
                continue;
            }
            if (same_pos && a.kind == EXCL_END && b.kind == INCL_START) {
                a.kind = EXCL_END_INCL_START;
                a.start_dt = b.end_dt;
                continue;
            }
        }
        out.push_back(std::move(u));
    }
    p.items = std::move(out);
    return p;
}

static std::vector<Partition> parse_all(const std::vector<std::string>& v) {
    std::vector<Partition> out;
    for (auto& s : v) out.push_back(parse_dsl(s));
    return out;
}

static Partition merge_all(const std::vector<Partition>& a, const std::vector<Partition>& b,
                           const Header& h) {
    std::vector<const Partition*> vs;
    for (auto& p : a) vs.push_back(&p);
    for (auto& p : b) vs.push_back(&p);
    if (vs.size() == 1) return *vs[0];
    return merge_partition_versions(vs, h);
}

// the test harness pipeline: merge data, merge tombstone sources, filter,
// purge (NOW=1000, GC_BEFORE=100 — no-ops on these vectors, run for fidelity)
static Partition compact_case(const std::vector<std::string>& inputs,
                              const std::vector<std::string>& tombs, const Header& h) {
    std::vector<Partition> in = parse_all(inputs), ts = parse_all(tombs);
    Partition data = merge_all(in, {}, h);
    Partition tomb = merge_all(ts, {}, h);
    garbage_filter(data, tomb, h, false);
    purge_partition(data, 1000, 100, false, {}, false);
    return data;
}

static bool items_equal(const Partition& a, const Partition& b) {
    if (!(a.del.mfda == b.del.mfda && a.del.ldt == b.del.ldt)) return false;
    if (a.items.size() != b.items.size()) return false;
    for (size_t i = 0; i < a.items.size(); i++) {
        const Unfiltered &x = a.items[i], &y = b.items[i];
        if (x.kind != y.kind) return false;
        if (x.kind == Unfiltered::ROW) {
            if (x.row.clustering[0].v != y.row.clustering[0].v) return false;
            if (x.row.live.ts != y.row.live.ts) return false;
            if (x.row.del.mfda != y.row.del.mfda || x.row.del.ldt != y.row.del.ldt) return false;
        } else {
            if (x.marker.kind != y.marker.kind) return false;
            if (x.marker.values[0].v != y.marker.values[0].v) return false;
            if (x.marker.end_dt.mfda != y.marker.end_dt.mfda) return false;
            if (x.marker.boundary() && x.marker.start_dt.mfda != y.marker.start_dt.mfda)
                return false;
        }
    }
    return true;
}

static int size_of(const Partition& p) {
    int n = 0;
    for (auto& u : p.items)
        n += (u.kind == Unfiltered::MARKER && u.marker.boundary()) ? 2 : 1;
    return n;
}

static void equivalence(const std::vector<std::string>& inputs,
                        const std::vector<std::string>& tombs, const Partition& result,
                        const Header& h, const char* name) {
    std::vector<Partition> in = parse_all(inputs), ts = parse_all(tombs);
    Partition lhs = merge_all(in, ts, h);
    std::vector<Partition> rs{result};
    Partition rhs = merge_all(rs, ts, h);
    std::string msg = std::string(name) + ": equivalence (merge(in+tombs) == merge(result+tombs))";
    CHECK(items_equal(lhs, rhs), msg.c_str());
}


static uint64_t gc_rnd(uint64_t& st) {
    st ^= st << 13;
    st ^= st >> 7;
    st ^= st << 17;
    return st;
}

// random VALID unfiltered stream in the DSL (normalized rows: deletion
// strictly below liveness — see the note at the call site)
static std::string gen_stream(uint64_t seed) {
    uint64_t st = seed * 2654435761u + 1;
    std::string out;
    int pos = (int)(gc_rnd(st) % 5);
    bool open = false;
    int64_t open_t = 0;
    for (int i = 0; i < 12 && pos < 200; i++) {
        int what = (int)(gc_rnd(st) % 3);
        int64_t t = 100 + (int)(gc_rnd(st) % 100);
        char buf[64];
        if (what == 0 && !open) {
            snprintf(buf, sizeof buf, "%d<%s[%lld] ", pos, gc_rnd(st) & 1 ? "=" : "",
                     (long long)t);
            open = true;
            open_t = t;
        } else if (what == 1 && open) {
            snprintf(buf, sizeof buf, "[%lld]<%s%d ", (long long)open_t,
                     gc_rnd(st) & 1 ? "=" : "", pos);
            open = false;
        } else {
            if (gc_rnd(st) % 4 == 0 && t > 101)
                snprintf(buf, sizeof buf, "%d[%lld D%lld] ", pos, (long long)t,
                         (long long)(100 + (int)(gc_rnd(st) % (t - 101))));
            else
                snprintf(buf, sizeof buf, "%d[%lld] ", pos, (long long)t);
        }
        std::string tokstr(buf);
        size_t sp = tokstr.find(" D");
        if (sp != std::string::npos) tokstr.erase(sp, 1);
        out += tokstr;
        pos += 1 + (int)(gc_rnd(st) % 9);
    }
    if (open) {
        char buf[64];
        snprintf(buf, sizeof buf, "[%lld]<%d ", (long long)open_t, pos);
        out += buf;
    }
    if (!out.empty()) out.pop_back();
    return out;
}

int main() {
    Header h = dsl_header();
    struct CountCase {
        const char* name;
        std::vector<std::string> in, ts;
        int max;
    };
    // CompactionIteratorTest.java:93-157
    std::vector<CountCase> cases = {
        {"SupersedeLeft", {"5<=[140] 10[150] [140]<20 22<[130] [130]<25 30[150]"},
         {"7<[160] 15[180] [160]<30 40[120]"}, 3},
        {"SupersedeMiddle", {"5<=[140] 10[150] [140]<40 60[150]"},
         {"7<=[160] 15[180] [160]<=30 40[120]"}, 3},
        {"SupersedeRight", {"9<=[140] 10[150] [140]<40 60[150]"},
         {"7<[160] 15[180] [160]<30 40[120]"}, 3},
        {"SwitchInSuperseded", {"5<=[140] 10[150] [140]<20 20<=[170] [170]<=50 60[150]"},
         {"7<[160] 15[180] [160]<30 40[120]"}, 5},
        {"Boundaries", {"5<=[120] [120]<9 9<=[140] 10[150] [140]<40 40<=[120] 60[150] [120]<90"},
         {"7<[160] 15[180] [160]<30 40[120] 45<[140] [140]<80 88<=[130] [130]<100"}, 7},
        {"Matches",
         {"5<=[120] [120]<=9 9<[140] 10[150] [140]<40 40<=[120] 60[150] [120]<90 120<=[100] [100]<130"},
         {"9<[160] 15[180] [160]<40 40[120] 45<[140] [140]<90 90<=[110] [110]<100 120<=[100] [100]<130"},
         5},
    };
    for (auto& c : cases) {
        Partition r = compact_case(c.in, c.ts, h);
        std::string msg = std::string("GcCompaction") + c.name + ": <= " +
                          std::to_string(c.max) + " unfiltereds (got " +
                          std::to_string(size_of(r)) + ")";
        CHECK(size_of(r) <= c.max, msg.c_str());
        equivalence(c.in, c.ts, r, h, c.name);
    }
    {   // testGcCompactionRowDeletion (CompactionIteratorTest.java:159-168)
        std::vector<std::string> in{"10[150] 20[160] 25[160] 30[170] 40[120] 50[120]"};
        std::vector<std::string> ts{
            "10<=[155] 20[200D180] 30[200D160] [155]<=30 40[150D130] 50[150D100]"};
        Partition r = compact_case(in, ts, h);
        Partition want = parse_dsl("25[160] 30[170] 50[120]");
        CHECK(items_equal(r, want), "GcCompactionRowDeletion: result == 25[160] 30[170] 50[120]");
        equivalence(in, ts, r, h, "RowDeletion");
    }
    {   // testGcCompactionPartitionDeletion (CompactionIteratorTest.java:170-181)
        std::vector<std::string> in{"10[150] 20[160] 25[160] 30[170] 40[120] 50[120]"};
        std::vector<std::string> ts{
            "D165|10<=[155] 20[200D180] 30[200D160] [155]<=30 40[150D130] 50[150D100]"};
        Partition r = compact_case(in, ts, h);
        Partition want = parse_dsl("30[170]");
        CHECK(items_equal(r, want), "GcCompactionPartitionDeletion: result == 30[170]");
        equivalence(in, ts, r, h, "PartitionDeletion");
    }
    // randomized equivalence sweep (testRandomGcCompaction analog,
    // CompactionIteratorTest.java:308-360): random valid unfiltered streams
    // through the same pipeline; the equivalence law and output validity
    // (strictly ordered, well-formed marker nesting) must hold for every seed
    {
        auto valid = [&](const Partition& p) {
            bool open = false;
            for (auto& u : p.items) {
                if (u.kind != Unfiltered::MARKER) continue;
                const Marker& mk = u.marker;
                if (mk.boundary()) {
                    if (!open) return false;
                } else if (mk.kind == INCL_START || mk.kind == EXCL_START) {
                    if (open) return false;
                    open = true;
                } else {
                    if (!open) return false;
                    open = false;
                }
            }
            return !open;
        };
        int bad = 0;
        for (uint64_t seed = 1; seed <= 300; seed++) {
            std::vector<std::string> in{gen_stream(seed * 3), gen_stream(seed * 3 + 1)};
            std::vector<std::string> ts{gen_stream(seed * 3 + 2)};
            Partition r = compact_case(in, ts, h);
            if (!valid(r)) {
                printf("FAIL seed %llu: invalid output marker sequence\n",
                       (unsigned long long)seed);
                bad++;
                continue;
            }
            std::vector<Partition> inp = parse_all(in), tsp = parse_all(ts);
            Partition lhs = merge_all(inp, tsp, h);
            std::vector<Partition> rs{r};
            Partition rhs = merge_all(rs, tsp, h);
            if (!items_equal(lhs, rhs)) {
                printf("FAIL seed %llu: equivalence broken\n", (unsigned long long)seed);
                bad++;
            }
        }
        char msg[96];
        snprintf(msg, sizeof msg, "random GC equivalence sweep: 300 seeds (%d bad)", bad);
        CHECK(bad == 0, msg);
        fails += bad ? 1 : 0;
    }
    // first-principles merge check (UnfilteredRowIteratorsMergeTest
    // testTombstoneMerge essence): merge N random streams, then verify the
    // output against a pointwise timeline oracle — at every sampled
    // clustering position (integers and midpoints), the merged active
    // deletion equals the max over sources, and each row survives with
    // max-liveness iff not shadowed (DeletionTime.deletes: mfda >= ts)
    {
        // active deletion of one partition at query q (in half-steps: q=2*ck
        // is the exact position, q=2*ck+1 is between ck and ck+1)
        auto active_at = [](const Partition& p, int q) {
            int64_t best = INT64_MIN;
            if (!p.del.live()) best = p.del.mfda;
            int64_t cur = INT64_MIN;
            for (auto& u : p.items) {
                if (u.kind != Unfiltered::MARKER) continue;
                const Marker& mk = u.marker;
                long long ck = 0;
                for (int i = 0; i < 8; i++) ck = (ck << 8) | mk.values[0].v[i];
                int mp = (int)(2 * ck);
                // evaluate BEFORE considering the marker if q < its bound;
                // bound position in half-steps per kind
                int bq;  // first q the post-marker state applies to
                if (mk.kind == INCL_START) bq = mp;
                else if (mk.kind == EXCL_START) bq = mp + 1;
                else if (mk.kind == INCL_END) bq = mp + 1;       // closes after ck
                else if (mk.kind == EXCL_END) bq = mp;           // closes before ck
                else if (mk.kind == EXCL_END_INCL_START) bq = mp;
                else bq = mp + 1;                                // INCL_END_EXCL_START
                if (q < bq) break;
                if (mk.boundary()) cur = mk.start_dt.mfda;
                else if (mk.open(false)) cur = mk.end_dt.mfda;
                else cur = INT64_MIN;
            }
            return std::max(best, cur);
        };
        int bad = 0;
        for (uint64_t seed = 1; seed <= 300 && bad < 5; seed++) {
            std::vector<std::string> in{gen_stream(seed * 7 + 1), gen_stream(seed * 7 + 2),
                                        gen_stream(seed * 7 + 3)};
            std::vector<Partition> ps = parse_all(in);
            Partition m = merge_all(ps, {}, h);
            for (int q = 0; q <= 2 * 210 && bad < 5; q++) {
                int64_t want = INT64_MIN;
                for (auto& p : ps) want = std::max(want, active_at(p, q));
                int64_t got = active_at(m, q);
                if (want != got) {
                    printf("FAIL seed %llu q=%d timeline: want %lld got %lld\n",
                           (unsigned long long)seed, q, (long long)want, (long long)got);
                    bad++;
                }
            }
            // row survival + liveness
            for (int ck = 0; ck <= 210 && bad < 5; ck++) {
                int64_t maxts = INT64_MIN, maxdel = INT64_MIN;
                for (auto& p : ps)
                    for (auto& u : p.items)
                        if (u.kind == Unfiltered::ROW) {
                            long long c = 0;
                            for (int i = 0; i < 8; i++) c = (c << 8) | u.row.clustering[0].v[i];
                            if (c != ck) continue;
                            maxts = std::max(maxts, u.row.live.ts);
                            if (!u.row.del.live()) maxdel = std::max(maxdel, u.row.del.mfda);
                        }
                int64_t act = INT64_MIN;
                for (auto& p : ps) act = std::max(act, active_at(p, 2 * ck));
                bool expect = (maxts != INT64_MIN && maxts > act) || maxdel > act;
                const Unfiltered* got = nullptr;
                for (auto& u : m.items)
                    if (u.kind == Unfiltered::ROW) {
                        long long c = 0;
                        for (int i = 0; i < 8; i++) c = (c << 8) | u.row.clustering[0].v[i];
                        if (c == ck) got = &u;
                    }
                if (expect != (got != nullptr)) {
                    printf("FAIL seed %llu ck=%d row survival: want %d\n",
                           (unsigned long long)seed, ck, (int)expect);
                    bad++;
                } else if (got && maxts > act && got->row.live.ts != maxts) {
                    printf("FAIL seed %llu ck=%d liveness: want %lld got %lld\n",
                           (unsigned long long)seed, ck, (long long)maxts,
                           (long long)got->row.live.ts);
                    bad++;
                }
            }
        }
        char msg[96];
        snprintf(msg, sizeof msg, "pointwise merge timeline oracle: 300 seeds (%d bad)", bad);
        CHECK(bad == 0, msg);
    }
    printf(fails ? "GC DSL vectors: %d FAILURES\n" : "GC DSL vectors: all OK\n", fails);
    return fails ? 1 : 0;
}
