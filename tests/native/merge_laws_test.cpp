// ORACLE law tests — merge/purge semantics transcribed from the reference's
// own unit tests (CompactionIteratorTest, UnfilteredRowIteratorsMergeTest,
// CompactionsPurgeTest, RowsTest/CellsTest tie rules) as direct assertions
// against oracle::merge_partition_versions / purge_partition. These pin the
// SEMANTICS the byte-level parity suite then carries to the GPU.
#include "../../oracle/src/compact.h"
#include "../../oracle/src/gen.h"
#include <cstdio>
#include <cstring>

using namespace oracle;

static int fails = 0;
#define CHECK(cond, msg)                                            \
    do {                                                            \
        if (!(cond)) { printf("FAIL: %s\n", msg); fails++; }        \
        else printf("ok:   %s\n", msg);                             \
    } while (0)

static Header simple_header() {
    Header h;
    h.key_type = CqlType::LONG;
    h.regular_cols = {{bytes{'v'}, CqlType::BYTES}};
    h.stats.min_ts = 1000;
    h.stats.min_ldt = 1000;
    h.stats.min_ttl = 0;
    return h;
}

static Partition part(int64_t tok) {
    Partition p;
    p.key = bytes{1, 2, 3};
    p.token = tok;
    return p;
}

static Row live_row(int64_t ts, const bytes& val) {
    Row r;
    r.live.ts = ts;
    r.cells.resize(1);
    Cell c;
    c.ts = ts;
    c.value = val;
    r.cells[0] = std::move(c);
    return r;
}

static Row tomb_row(int64_t ts, uint32_t ldt) {
    Row r;
    r.del.mfda = ts;
    r.del.ldt = ldt;
    r.cells.resize(1);
    return r;
}

static Unfiltered urow(Row r) {
    Unfiltered u;
    u.kind = Unfiltered::ROW;
    u.row = std::move(r);
    return u;
}

int main() {
    Header h = simple_header();

    // ---- Cells.reconcile: higher timestamp wins (Cells.java:145-151) ----
    {
        Partition a = part(7), b = part(7);
        a.items.push_back(urow(live_row(2000, bytes{'A'})));
        b.items.push_back(urow(live_row(3000, bytes{'B'})));
        Partition m = merge_partition_versions({&a, &b}, h);
        CHECK(m.items.size() == 1 && m.items[0].row.cells[0]->value == bytes{'B'} &&
                  m.items[0].row.cells[0]->ts == 3000,
              "cell reconcile: higher timestamp wins");
    }
    // ---- equal ts: tombstone cell beats live cell (Cells.java:153-161) ----
    {
        Partition a = part(7), b = part(7);
        a.items.push_back(urow(live_row(2000, bytes{'A'})));
        Row t;
        t.live.ts = 2000;
        t.cells.resize(1);
        Cell dc;
        dc.ts = 2000;
        dc.ldt = 1500;  // deleted cell (empty value)
        t.cells[0] = std::move(dc);
        b.items.push_back(urow(std::move(t)));
        Partition m = merge_partition_versions({&a, &b}, h);
        CHECK(m.items.size() == 1 && m.items[0].row.cells[0] &&
                  m.items[0].row.cells[0]->ldt == 1500,
              "cell reconcile: equal ts, tombstone cell beats live cell");
    }
    // ---- equal ts, both live: lexicographically greater value wins ----
    {
        Partition a = part(7), b = part(7);
        a.items.push_back(urow(live_row(2000, bytes{'Z', '1'})));
        b.items.push_back(urow(live_row(2000, bytes{'A', '2'})));
        Partition m = merge_partition_versions({&a, &b}, h);
        CHECK(m.items[0].row.cells[0]->value == (bytes{'Z', '1'}),
              "cell reconcile: equal ts ties break to greater value bytes");
    }
    // ---- row deletion shadows older cells, not newer (Row.Merger) ----
    {
        Partition a = part(7), b = part(7);
        a.items.push_back(urow(live_row(2000, bytes{'A'})));
        b.items.push_back(urow(tomb_row(2500, 1600)));
        Partition m = merge_partition_versions({&a, &b}, h);
        CHECK(m.items.size() == 1 && !m.items[0].row.cells[0] &&
                  m.items[0].row.del.mfda == 2500,
              "row deletion (ts 2500) shadows older cell (ts 2000)");
        Partition c = part(7), d = part(7);
        c.items.push_back(urow(live_row(3000, bytes{'A'})));
        d.items.push_back(urow(tomb_row(2500, 1600)));
        Partition m2 = merge_partition_versions({&c, &d}, h);
        CHECK(m2.items.size() == 1 && m2.items[0].row.cells[0] &&
                  m2.items[0].row.cells[0]->ts == 3000,
              "row deletion does NOT shadow newer cell (ts 3000 > 2500)");
    }
    // ---- partition deletion shadows older rows across versions ----
    {
        Partition a = part(7), b = part(7);
        a.items.push_back(urow(live_row(2000, bytes{'A'})));
        b.del = DeletionTime{2600, 1600};
        Partition m = merge_partition_versions({&a, &b}, h);
        CHECK(m.items.empty() && m.del.mfda == 2600,
              "partition deletion (ts 2600) shadows older row (ts 2000)");
    }
    // ---- LivenessInfo.supersedes: equal ts, expiring beats non-expiring ----
    {
        Partition a = part(7), b = part(7);
        Row r1;
        r1.live.ts = 2000;
        r1.cells.resize(1);
        a.items.push_back(urow(std::move(r1)));
        Row r2;
        r2.live.ts = 2000;
        r2.live.ttl = 100;
        r2.live.let = 5000;
        r2.cells.resize(1);
        b.items.push_back(urow(std::move(r2)));
        Partition m = merge_partition_versions({&a, &b}, h);
        CHECK(m.items[0].row.live.ttl == 100,
              "liveness merge: equal ts, expiring supersedes non-expiring");
    }
    // ---- purge: gcBefore drops expired tombstones; overlap table gates ----
    {
        Partition p = part(7);
        p.items.push_back(urow(tomb_row(2000, 1400)));
        Partition q = p;
        bool kept = purge_partition(q, 1800000000, /*gc_before=*/1500, false, {}, false);
        CHECK(!kept, "purge: row tombstone ldt 1400 < gcBefore 1500 purged (no overlaps)");
        Partition q2 = p;
        std::vector<PurgeRange> ov = {{INT64_MIN, INT64_MAX, /*min_ts=*/1000}};
        kept = purge_partition(q2, 1800000000, 1500, false, ov, false);
        CHECK(kept && q2.items.size() == 1,
              "purge: overlap min_ts 1000 <= tombstone ts 2000 blocks the purge");
        Partition q3 = p;
        std::vector<PurgeRange> ov2 = {{INT64_MIN, INT64_MAX, /*min_ts=*/2500}};
        kept = purge_partition(q3, 1800000000, 1500, false, ov2, false);
        CHECK(!kept, "purge: overlap min_ts 2500 > tombstone ts 2000 allows the purge");
    }
    // ---- range tombstone shadows rows in range after merge ----
    {
        Header hc = simple_header();
        hc.clustering_types = {CqlType::LONG};
        auto be8 = [](int64_t v) {
            bytes b(8);
            for (int i = 0; i < 8; i++) b[i] = (uint8_t)((uint64_t)v >> (8 * (7 - i)));
            return b;
        };
        auto ckrow = [&](int64_t ck, int64_t ts) {
            Row r = live_row(ts, bytes{'V'});
            r.clustering = {ClusterVal{ClusterVal::VALUE, be8(ck)}};
            return urow(std::move(r));
        };
        Partition a = part(7), b = part(7);
        a.items.push_back(ckrow(10, 2000));
        a.items.push_back(ckrow(20, 2000));
        a.items.push_back(ckrow(30, 2000));
        Unfiltered open_m;
        open_m.kind = Unfiltered::MARKER;
        open_m.marker.kind = INCL_START;
        open_m.marker.values = {ClusterVal{ClusterVal::VALUE, be8(15)}};
        open_m.marker.end_dt = DeletionTime{2500, 1600};
        Unfiltered close_m;
        close_m.kind = Unfiltered::MARKER;
        close_m.marker.kind = INCL_END;
        close_m.marker.values = {ClusterVal{ClusterVal::VALUE, be8(25)}};
        close_m.marker.end_dt = DeletionTime{2500, 1600};
        b.items.push_back(std::move(open_m));
        b.items.push_back(std::move(close_m));
        Partition m = merge_partition_versions({&a, &b}, h.clustering_types.empty() ? hc : hc);
        int rows = 0, markers = 0;
        for (auto& u : m.items) (u.kind == Unfiltered::ROW ? rows : markers)++;
        CHECK(rows == 2 && markers == 2,
              "range tombstone [15,25]@2500 shadows row ck=20@2000, keeps 10 and 30");
    }
    // ---- GarbageSkipper laws (CompactionIterator.java:401-598 / GcCompactionTest) ----
    {
        // source partition deletion shadows older data; the deletion itself
        // is NOT copied to the output
        Partition d = part(7), t = part(7);
        d.items.push_back(urow(live_row(2000, bytes{'A'})));
        t.del = DeletionTime{2500, 1600};
        garbage_filter(d, t, h, false);
        CHECK(d.items.empty() && d.del.live(),
              "gc: source partition deletion removes shadowed data, is not copied");
    }
    {
        // data newer than the source deletion survives
        Partition d = part(7), t = part(7);
        d.items.push_back(urow(live_row(3000, bytes{'A'})));
        t.del = DeletionTime{2500, 1600};
        garbage_filter(d, t, h, false);
        CHECK(d.items.size() == 1, "gc: data newer than source deletion survives");
    }
    {
        // source row deletion (row-level mode) removes the matching row only
        Partition d = part(7), t = part(7);
        d.items.push_back(urow(live_row(2000, bytes{'A'})));
        t.items.push_back(urow(tomb_row(2500, 1600)));
        garbage_filter(d, t, h, false);
        CHECK(d.items.empty(), "gc row-level: source row deletion removes shadowed row");
    }
    {
        // cell-level: a NEWER source cell overwrites (removes) the data cell;
        // an older one does not
        Partition d = part(7), t = part(7);
        d.items.push_back(urow(live_row(2000, bytes{'A'})));
        t.items.push_back(urow(live_row(2600, bytes{'B'})));
        garbage_filter(d, t, h, true);
        // the overwritten cell is removed; the pk liveness (row marker)
        // survives (Rows.removeShadowedCells keeps undeleted liveness)
        CHECK(d.items.size() == 1 && !d.items[0].row.cells[0] &&
                  !d.items[0].row.live.empty(),
              "gc cell-level: newer source cell removes data cell, keeps liveness");
        Partition d2 = part(7), t2 = part(7);
        d2.items.push_back(urow(live_row(2000, bytes{'A'})));
        t2.items.push_back(urow(live_row(1500, bytes{'B'})));
        garbage_filter(d2, t2, h, true);
        CHECK(d2.items.size() == 1, "gc cell-level: older source cell keeps data cell");
        // row-level mode ignores live source cells entirely
        Partition d3 = part(7), t3 = part(7);
        d3.items.push_back(urow(live_row(2000, bytes{'A'})));
        t3.items.push_back(urow(live_row(2600, bytes{'B'})));
        garbage_filter(d3, t3, h, false);
        CHECK(d3.items.size() == 1, "gc row-level: live source cells do not shadow");
    }
    {
        // source range tombstone removes covered data rows; the range itself
        // is not copied
        Header hc = simple_header();
        hc.clustering_types = {CqlType::LONG};
        auto be8 = [](int64_t v) {
            bytes b(8);
            for (int i = 0; i < 8; i++) b[i] = (uint8_t)((uint64_t)v >> (8 * (7 - i)));
            return b;
        };
        auto ckrow = [&](int64_t ck, int64_t ts) {
            Row r = live_row(ts, bytes{'V'});
            r.clustering = {ClusterVal{ClusterVal::VALUE, be8(ck)}};
            return urow(std::move(r));
        };
        Partition d = part(7), t = part(7);
        d.items.push_back(ckrow(10, 2000));
        d.items.push_back(ckrow(20, 2000));
        d.items.push_back(ckrow(30, 2000));
        Unfiltered o, c;
        o.kind = Unfiltered::MARKER;
        o.marker.kind = INCL_START;
        o.marker.values = {ClusterVal{ClusterVal::VALUE, be8(15)}};
        o.marker.end_dt = DeletionTime{2500, 1600};
        c.kind = Unfiltered::MARKER;
        c.marker.kind = INCL_END;
        c.marker.values = {ClusterVal{ClusterVal::VALUE, be8(25)}};
        c.marker.end_dt = DeletionTime{2500, 1600};
        t.items.push_back(std::move(o));
        t.items.push_back(std::move(c));
        garbage_filter(d, t, hc, false);
        int rows = 0, markers = 0;
        for (auto& u : d.items) (u.kind == Unfiltered::ROW ? rows : markers)++;
        CHECK(rows == 2 && markers == 0,
              "gc: source range tombstone removes covered rows, copies nothing");
    }
    printf(fails ? "FAILED %d\n" : "merge laws OK\n", fails);
    return fails ? 1 : 0;
}
