// ORACLE — test infrastructure only (see util.h header note).
// CPU restatement of the CompactionIterator merge pipeline:
//   UnfilteredPartitionIterators.merge (UnfilteredPartitionIterators.java:123-218)
//   UnfilteredRowIterators.merge      (UnfilteredRowIterators.java:400-599)
//   Row.Merger / ColumnDataReducer    (Row.java:694-860)
//   Cells.reconcile                   (Cells.java:68-119)
//   RangeTombstoneMarker.Merger       (RangeTombstoneMarker.java:72-198)
//   PurgeFunction                     (PurgeFunction.java:26-145)
#include "compact.h"
#include <algorithm>
#include <functional>

namespace oracle {

// ---------------------------------------------------------------------------
// comparator (ClusteringComparator.java:140-158 + ClusteringPrefix.Kind)
// ---------------------------------------------------------------------------
static int kind_comparison(BoundKind k) {
    static const int tbl[8] = {0, 0, 0, 1, 2, 3, 3, 3};  // Kind(comparison, …) ctor args
    return tbl[k];
}
static int kind_compared_to_clustering(BoundKind k) {
    static const int tbl[8] = {-1, -1, -1, -1, 0, 1, 1, 1};
    return tbl[k];
}
static int compare_component(const Header& h, size_t i, const ClusterVal& a, const ClusterVal& b) {
    if (a.state == ClusterVal::NUL) return b.state == ClusterVal::NUL ? 0 : -1;
    if (b.state == ClusterVal::NUL) return 1;
    const bytes ea, eb;
    const bytes& va = a.state == ClusterVal::EMPTY ? ea : a.v;
    const bytes& vb = b.state == ClusterVal::EMPTY ? eb : b.v;
    return compare_typed(h.clustering_types[i], va, vb);
}
int compare_clustering_prefix(const Header& h, BoundKind ka, const Clustering& a,
                              BoundKind kb, const Clustering& b) {
    size_t mn = std::min(a.size(), b.size());
    for (size_t i = 0; i < mn; i++) {
        int c = compare_component(h, i, a[i], b[i]);
        if (c) return c;
    }
    if (a.size() == b.size()) {
        int c1 = kind_comparison(ka), c2 = kind_comparison(kb);
        return c1 < c2 ? -1 : c1 > c2 ? 1 : 0;
    }
    return a.size() < b.size() ? kind_compared_to_clustering(ka) : -kind_compared_to_clustering(kb);
}
static int compare_unfiltered(const Header& h, const Unfiltered& a, const Unfiltered& b) {
    BoundKind ka = a.kind == Unfiltered::ROW ? CLUSTERING_K : a.marker.kind;
    BoundKind kb = b.kind == Unfiltered::ROW ? CLUSTERING_K : b.marker.kind;
    return compare_clustering_prefix(h, ka, a.clustering(), kb, b.clustering());
}

// ---------------------------------------------------------------------------
// Cells.reconcile (Cells.java:68-119)
// ---------------------------------------------------------------------------
static const Cell& reconcile_cells(const Cell& left, const Cell& right) {
    if (left.ts != right.ts) return left.ts > right.ts ? left : right;
    bool l_dt = left.ldt != LDT_NONE_U32, r_dt = right.ldt != LDT_NONE_U32;
    if (l_dt | r_dt) {
        if (l_dt != r_dt) return l_dt ? left : right;
        bool l_tomb = !left.expiring(), r_tomb = !right.expiring();
        if (l_tomb != r_tomb) return l_tomb ? left : right;
        if (left.ldt != right.ldt)
            return ldt_to_long(left.ldt) > ldt_to_long(right.ldt) ? left : right;
    }
    // compareValues: unsigned lexicographic on value bytes (ValueAccessor.compare)
    size_t n = std::min(left.value.size(), right.value.size());
    int c = memcmp(left.value.data(), right.value.data(), n);
    if (c == 0) c = left.value.size() == right.value.size() ? 0 : (left.value.size() < right.value.size() ? -1 : 1);
    return c >= 0 ? left : right;
}

// ---------------------------------------------------------------------------
// Repair-validation digest (repair/Validator.rowHash:207-216 +
// UnfilteredRowIterators.digest + Rows/Cells/DeletionTime/ClusteringPrefix
// .digest): concat(murmur3_128(seed 1000), murmur3_128(seed 2000)) over the
// partition's field stream — big-endian ints/longs, INVERTED booleans
// (Digest.updateWithBoolean), counter contexts digested body-only
// (Digest.updateWithCounterContext skips the header), cell
// localDeletionTime deliberately excluded (DeletionTime.digest).
// ---------------------------------------------------------------------------
void validator_digest(const Partition& p, const Header& h, uint8_t out[32]) {
    M3Stream a, b;
    a.init(1000);
    b.init(2000);
    auto put = [&](const uint8_t* d, size_t n) { a.put(d, n); b.put(d, n); };
    auto u8 = [&](uint8_t v) { a.put_u8(v); b.put_u8(v); };
    auto i32 = [&](int32_t v) { a.put_i32be(v); b.put_i32be(v); };
    auto i64 = [&](int64_t v) { a.put_i64be(v); b.put_i64be(v); };
    auto bl = [&](bool v) { a.put_bool(v); b.put_bool(v); };
    auto put_cell = [&](const Cell& c, bool counter_col, const bytes* path) {
        bool is_counter_cell = counter_col && c.ldt == LDT_NONE_U32;
        if (is_counter_cell) {
            if (!c.value.empty()) {
                int16_t hn = (int16_t)((c.value[0] << 8) | c.value[1]);
                size_t hl = 2 + (size_t)(hn < 0 ? -hn : hn) * 2;
                if (c.value.size() > hl) put(c.value.data() + hl, c.value.size() - hl);
            }
        } else {
            put(c.value.data(), c.value.size());
        }
        i64(c.ts);
        i32(c.ttl == NO_TTL ? 0 : c.ttl);
        bl(is_counter_cell);
        if (path) put(path->data(), path->size());
    };
    auto digest_row = [&](const Row& r, bool is_static) {
        u8(0);  // Unfiltered.Kind.ROW
        if (!is_static)
            for (auto& cv : r.clustering) put(cv.v.data(), cv.v.size());
        u8(is_static ? 3 : 4);  // ClusteringPrefix.Kind ordinal
        i64(r.del.mfda);
        bl(false);  // Row.Deletion.isShadowable
        i64(r.live.ts);
        const auto& cols = is_static ? h.static_cols : h.regular_cols;
        for (size_t ci = 0; ci < cols.size(); ci++) {
            if (is_complex_type(cols[ci].second)) {
                if (ci >= r.complex.size() || !r.complex[ci]) continue;
                const ComplexData& cd = *r.complex[ci];
                if (!cd.del.live()) i64(cd.del.mfda);
                for (const Cell& c : cd.cells) put_cell(c, false, &c.path);
            } else if (ci < r.cells.size() && r.cells[ci]) {
                put_cell(*r.cells[ci], is_counter_type(cols[ci].second), nullptr);
            }
        }
    };
    put(p.key.data(), p.key.size());
    i64(p.del.mfda);
    for (auto& [nm, t2] : h.regular_cols) { (void)t2; put(nm.data(), nm.size()); }
    bool static_present = h.has_static() && !row_is_empty(p.static_row);
    if (static_present)
        for (auto& [nm, t2] : h.static_cols) { (void)t2; put(nm.data(), nm.size()); }
    bl(false);  // isReverseOrder
    if (static_present) {
        Row sr = p.static_row;
        digest_row(sr, true);
    } else {
        Row empty;  // Rows.EMPTY_STATIC_ROW
        digest_row(empty, true);
    }
    for (const Unfiltered& u : p.items) {
        if (u.kind == Unfiltered::ROW) {
            digest_row(u.row, false);
        } else {
            const Marker& m = u.marker;
            u8(1);  // Unfiltered.Kind.RANGE_TOMBSTONE_MARKER
            for (auto& cv : m.values) put(cv.v.data(), cv.v.size());
            u8((uint8_t)m.kind);
            i64(m.end_dt.mfda);
            if (m.boundary()) i64(m.start_dt.mfda);
        }
    }
    a.final16(out);
    b.final16(out + 16);
}

// ---------------------------------------------------------------------------
// CounterContext (db/context/CounterContext.java): header of global/local
// element flags + (CounterId[16], clock i64, count i64) shards in unsigned
// id order. merge() transcribed pairwise (compare(): global beats all with
// (clock, count) ties; local+local sums (DISJOINT); remote by the legacy-
// aware clock rule then count). Cell format fixture-pinned by the
// reference's legacy_oa_*_counter sstables.
// ---------------------------------------------------------------------------
struct CtxShard {
    uint8_t id[16];
    int64_t clock, count;
    bool global = false, local = false;
};
static std::vector<CtxShard> ctx_parse(const bytes& v) {
    std::vector<CtxShard> out;
    if (v.size() < 2) return out;
    int16_t hdr = (int16_t)((v[0] << 8) | v[1]);
    int n_flagged = hdr < 0 ? -hdr : hdr;
    size_t body = 2 + (size_t)n_flagged * 2;
    size_t nsh = (v.size() - body) / 32;
    std::vector<int16_t> elts(n_flagged);
    for (int i = 0; i < n_flagged; i++) elts[i] = (int16_t)((v[2 + 2 * i] << 8) | v[3 + 2 * i]);
    for (size_t i = 0; i < nsh; i++) {
        CtxShard sh;
        memcpy(sh.id, v.data() + body + i * 32, 16);
        uint64_t c1 = 0, c2 = 0;
        for (int b = 0; b < 8; b++) c1 = (c1 << 8) | v[body + i * 32 + 16 + b];
        for (int b = 0; b < 8; b++) c2 = (c2 << 8) | v[body + i * 32 + 24 + b];
        sh.clock = (int64_t)c1;
        sh.count = (int64_t)c2;
        for (int16_t e : elts) {
            if (hdr >= 0 && e == (int16_t)(i + INT16_MIN)) sh.global = true;
            if (e == (int16_t)i && hdr >= 0) sh.local = true;
            if (hdr < 0 && e == (int16_t)i) sh.local = true;  // pre-2.1 header
        }
        out.push_back(sh);
    }
    return out;
}
static bytes ctx_serialize(const std::vector<CtxShard>& shards) {
    bytes out;
    std::vector<int16_t> elts;
    for (size_t i = 0; i < shards.size(); i++) {
        if (shards[i].global) elts.push_back((int16_t)(i + INT16_MIN));
        else if (shards[i].local) elts.push_back((int16_t)i);
    }
    out.push_back((uint8_t)(elts.size() >> 8));
    out.push_back((uint8_t)elts.size());
    for (int16_t e : elts) {
        out.push_back((uint8_t)((uint16_t)e >> 8));
        out.push_back((uint8_t)e);
    }
    for (auto& sh : shards) {
        out.insert(out.end(), sh.id, sh.id + 16);
        for (int b = 7; b >= 0; b--) out.push_back((uint8_t)((uint64_t)sh.clock >> (8 * b)));
        for (int b = 7; b >= 0; b--) out.push_back((uint8_t)((uint64_t)sh.count >> (8 * b)));
    }
    return out;
}
// compare(): -1 LESS, 0 EQUAL, +1 GREATER, 2 DISJOINT (CounterContext.java:~300)
static int ctx_compare(const CtxShard& l, const CtxShard& r) {
    if (l.global || r.global) {
        if (l.global && r.global) {
            if (l.clock == r.clock)
                return l.count > r.count ? 1 : l.count == r.count ? 0 : -1;
            return l.clock > r.clock ? 1 : -1;
        }
        return l.global ? 1 : -1;
    }
    if (l.local || r.local) {
        if (l.local && r.local) return 2;
        return l.local ? 1 : -1;
    }
    if (l.clock == r.clock)
        return l.count > r.count ? 1 : l.count == r.count ? 0 : -1;
    if ((l.clock >= 0 && r.clock > 0 && l.clock >= r.clock) ||
        (l.clock < 0 && (r.clock > 0 || l.clock < r.clock)))
        return 1;
    return -1;
}
static bytes ctx_merge(const bytes& lv, const bytes& rv) {
    auto L = ctx_parse(lv), R = ctx_parse(rv);
    std::vector<CtxShard> out;
    size_t i = 0, j = 0;
    bool lsup = true, rsup = true;
    while (i < L.size() && j < R.size()) {
        int c = memcmp(L[i].id, R[j].id, 16);
        if (c == 0) {
            int rel = ctx_compare(L[i], R[j]);
            if (rel == 2) {  // DISJOINT: two local shards sum
                CtxShard m = L[i];
                m.clock += R[j].clock;
                m.count += R[j].count;
                out.push_back(m);
                lsup = rsup = false;
            } else if (rel > 0) {
                out.push_back(L[i]);
                rsup = false;
            } else {
                out.push_back(R[j]);
                if (rel < 0) lsup = false;
            }
            i++; j++;
        } else if (c < 0) {
            out.push_back(L[i++]);
            rsup = false;
        } else {
            out.push_back(R[j++]);
            lsup = false;
        }
    }
    if (i < L.size()) rsup = false;
    if (j < R.size()) lsup = false;
    while (i < L.size()) out.push_back(L[i++]);
    while (j < R.size()) out.push_back(R[j++]);
    // superset early-return keeps the side's exact bytes (merge():~)
    if (lsup) return lv;
    if (rsup) return rv;
    return ctx_serialize(out);
}
// Cells.resolveCounter (Cells.java:121-162): tombstone beats any live
// counter; else context merge with ts = max. Deliberate divergence: the
// reference lets an EMPTY value win a pairwise merge — empty counter cells
// are a read-path artifact (#10657/#11726) that is never serialized into an
// sstable, so this branch is unreachable for compaction inputs; both this
// oracle and the GPU fold use the simpler non-empty-wins rule there.
static Cell counter_reconcile(const Cell& l, const Cell& r) {
    bool lt = l.ldt != LDT_NONE_U32, rt = r.ldt != LDT_NONE_U32;
    if (lt || rt) {
        if (lt != rt) return lt ? l : r;
        // two tombstones: regular rules (both land in resolveRegular)
        return reconcile_cells(l, r);
    }
    bool le = l.value.empty(), re2 = r.value.empty();
    if (le || re2) {
        if (le != re2) return le ? r : l;
        return l.ts > r.ts ? l : r;
    }
    Cell out = l;
    out.value = ctx_merge(l.value, r.value);
    out.ts = std::max(l.ts, r.ts);
    out.ldt = LDT_NONE_U32;
    out.ttl = NO_TTL;
    return out;
}

// ---------------------------------------------------------------------------
// Row.Merger.merge (Row.java:730-781) — simple columns only
// ---------------------------------------------------------------------------
static bool row_merge(const std::vector<const Row*>& versions, const DeletionTime& active_deletion,
                      const std::vector<std::pair<bytes, CqlType>>& cols, Row& out) {
    const size_t ncols = cols.size();
    // returns false if merged row is null (fully shadowed/empty)
    int present = 0;
    const Row* last = nullptr;
    for (auto* r : versions) if (r) { present++; last = r; }
    if (present == 1 && active_deletion.live()) { out = *last; return true; }

    LivenessInfo info;          // LivenessInfo.EMPTY
    DeletionTime row_del;       // Deletion.LIVE
    const Clustering* clust = nullptr;
    for (auto* r : versions) {
        if (!r) continue;
        clust = &r->clustering;
        if (r->live.supersedes(info)) info = r->live;
        if (r->del.supersedes(row_del)) row_del = r->del;
    }
    // (rowDeletion.isShadowedBy(rowInfo) only applies to deprecated shadowable
    //  deletions — always false for oa-era data, Row.java:430-433)
    DeletionTime active = active_deletion;
    if (row_del.supersedes(active)) active = row_del;
    else row_del = DT_LIVE;
    if (active.deletes(info.ts)) info = LivenessInfo{};

    out = Row{};
    out.clustering = clust ? *clust : Clustering{};
    out.live = info;
    out.del = row_del;
    out.cells.assign(ncols, std::nullopt);
    bool any_cpx_col = false;
    for (auto& cp : cols) any_cpx_col |= is_complex_type(cp.second);
    if (any_cpx_col) out.complex.assign(ncols, std::nullopt);
    bool any_cell = false;
    for (size_t ci = 0; ci < ncols; ci++) {
        if (is_complex_type(cols[ci].second)) {
            // ColumnDataReducer complex branch (Row.java:851-884)
            std::vector<const ComplexData*> cds;
            for (auto* r : versions)
                if (r && ci < r->complex.size() && r->complex[ci]) cds.push_back(&*r->complex[ci]);
            if (cds.empty()) continue;
            DeletionTime complex_del;  // LIVE
            for (auto* cd : cds)
                if (cd->del.supersedes(complex_del)) complex_del = cd->del;
            ComplexData out_cd;
            DeletionTime cell_active = active;
            if (complex_del.supersedes(active)) {
                cell_active = complex_del;
                out_cd.del = complex_del;  // addComplexDeletion
            }
            // k-way path merge with CellReducer (Row.java:893-909): per path,
            // versions reduce in source order, cells deleted by cell_active skip
            std::vector<size_t> cpos(cds.size(), 0);
            while (true) {
                const bytes* minp = nullptr;
                for (size_t i = 0; i < cds.size(); i++) {
                    if (cpos[i] >= cds[i]->cells.size()) continue;
                    const bytes& pth = cds[i]->cells[cpos[i]].path;
                    if (!minp || compare_cell_path(pth, *minp) < 0) minp = &pth;
                }
                if (!minp) break;
                const Cell* merged = nullptr;
                for (size_t i = 0; i < cds.size(); i++) {
                    if (cpos[i] >= cds[i]->cells.size()) continue;
                    const Cell& cell = cds[i]->cells[cpos[i]];
                    if (compare_cell_path(cell.path, *minp) != 0) continue;
                    cpos[i]++;
                    if (cell_active.deletes(cell.ts)) continue;
                    merged = merged == nullptr ? &cell : &reconcile_cells(*merged, cell);
                }
                if (merged) out_cd.cells.push_back(*merged);
            }
            // Builder.build: live deletion + no cells -> null column
            if (out_cd.del.live() && out_cd.cells.empty()) continue;
            out.complex[ci] = std::move(out_cd);
            any_cell = true;
            continue;
        }
        if (is_counter_type(cols[ci].second)) {
            // Cells.resolveCounter chain (value-semantics: merged contexts
            // allocate fresh bytes)
            std::optional<Cell> mc;
            for (auto* r : versions) {
                if (!r || ci >= r->cells.size() || !r->cells[ci]) continue;
                const Cell& cell = *r->cells[ci];
                if (active.deletes(cell.ts)) continue;
                mc = mc ? counter_reconcile(*mc, cell) : cell;
            }
            if (mc) { out.cells[ci] = std::move(*mc); any_cell = true; }
            continue;
        }
        const Cell* merged = nullptr;
        for (auto* r : versions) {  // version order == source order (reduce call order)
            if (!r || ci >= r->cells.size() || !r->cells[ci]) continue;
            const Cell& cell = *r->cells[ci];
            if (active.deletes(cell.ts)) continue;  // ColumnDataReducer.getReduced (Row.java:~840)
            merged = merged == nullptr ? &cell : &reconcile_cells(*merged, cell);
        }
        if (merged) { out.cells[ci] = *merged; any_cell = true; }
    }
    if (info.empty() && row_del.live() && !any_cell) return false;  // null row
    return true;
}

// ---------------------------------------------------------------------------
// RangeTombstoneMarker.Merger (RangeTombstoneMarker.java:72-198)
// ---------------------------------------------------------------------------
struct MarkerMerger {
    DeletionTime partition_deletion;
    std::vector<std::optional<DeletionTime>> open_markers;  // per source
    int biggest = -1;

    explicit MarkerMerger(size_t k, const DeletionTime& pd)
        : partition_deletion(pd), open_markers(k) {}

    DeletionTime current_open_in_merged() const {
        if (biggest < 0) return DT_LIVE;
        const DeletionTime& d = *open_markers[biggest];
        return !d.supersedes(partition_deletion) ? DT_LIVE : d;
    }
    DeletionTime active_deletion() const {
        DeletionTime om = current_open_in_merged();
        return om.live() ? partition_deletion : om;
    }
    // versions: per-source marker at this position (or null). Returns merged
    // marker or nullopt (no marker emitted).
    std::optional<Marker> merge(const std::vector<const Marker*>& versions) {
        DeletionTime prev = current_open_in_merged();
        const Marker* bound_src = nullptr;
        for (size_t i = 0; i < versions.size(); i++) {
            const Marker* m = versions[i];
            if (!m) continue;
            bound_src = m;  // `bound` = last add()ed marker's clustering
            if (m->open(false)) open_markers[i] = m->open_dt();
            else open_markers[i] = std::nullopt;
        }
        biggest = -1;
        for (size_t i = 0; i < open_markers.size(); i++)
            if (open_markers[i] && (biggest < 0 || open_markers[i]->supersedes(*open_markers[biggest])))
                biggest = (int)i;
        DeletionTime next = current_open_in_merged();
        if (prev == next) return std::nullopt;
        bool before_clustering = kind_compared_to_clustering(bound_src->kind) < 0;
        Marker m;
        m.values = bound_src->values;
        if (prev.live()) {
            m.kind = before_clustering ? INCL_START : EXCL_START;
            m.end_dt = next;
        } else if (next.live()) {
            m.kind = before_clustering ? EXCL_END : INCL_END;
            m.end_dt = prev;
        } else {
            m.kind = before_clustering ? EXCL_END_INCL_START : INCL_END_EXCL_START;
            m.end_dt = prev;
            m.start_dt = next;
        }
        return m;
    }
};

// ---------------------------------------------------------------------------
// partition-version merge (UnfilteredRowIterators.merge)
// ---------------------------------------------------------------------------
Partition merge_partition_versions(const std::vector<const Partition*>& versions, const Header& h) {
    if (versions.size() == 1) return *versions[0];  // merge(List size 1) returns as-is
    Partition out;
    out.key = versions[0]->key;
    out.token = versions[0]->token;
    // collectPartitionLevelDeletion (UnfilteredRowIterators.java:465-482)
    DeletionTime del;
    for (auto* p : versions)
        if (!del.supersedes(p->del)) del = p->del;
    out.del = del;

    // static rows merge like rows with activeDeletion = the merged partition
    // deletion (no range tombstone can cover the static clustering)
    if (h.has_static()) {
        std::vector<const Row*> svs;
        for (auto* p : versions)
            if (!row_is_empty(p->static_row)) svs.push_back(&p->static_row);
        if (!svs.empty()) {
            Row merged_static;
            if (row_merge(svs, del, h.static_cols, merged_static))
                out.static_row = std::move(merged_static);
            out.static_row.static_flag = true;
        }
    }

    size_t k = versions.size();
    std::vector<size_t> pos(k, 0);
    MarkerMerger marker_merger(k, del);
    const auto& rcols = h.regular_cols;
    while (true) {
        // find min position among streams
        int min_src = -1;
        for (size_t i = 0; i < k; i++) {
            if (pos[i] >= versions[i]->items.size()) continue;
            if (min_src < 0 ||
                compare_unfiltered(h, versions[i]->items[pos[i]], versions[min_src]->items[pos[min_src]]) < 0)
                min_src = (int)i;
        }
        if (min_src < 0) break;
        const Unfiltered& first = versions[min_src]->items[pos[min_src]];
        // gather all sources equal to min — reduce() call order is source order
        std::vector<const Row*> row_versions(k, nullptr);
        std::vector<const Marker*> marker_versions(k, nullptr);
        bool is_row = first.kind == Unfiltered::ROW;
        for (size_t i = 0; i < k; i++) {
            if (pos[i] >= versions[i]->items.size()) continue;
            const Unfiltered& u = versions[i]->items[pos[i]];
            if (compare_unfiltered(h, u, first) != 0) continue;
            // same position: rows group with rows, markers with markers (kind
            // comparison separates CLUSTERING from bounds)
            if (u.kind == Unfiltered::ROW) row_versions[i] = &u.row;
            else marker_versions[i] = &u.marker;
            pos[i]++;
        }
        if (is_row) {
            Row merged;
            if (row_merge({row_versions.begin(), row_versions.end()}, marker_merger.active_deletion(), rcols, merged)) {
                Unfiltered u;
                u.kind = Unfiltered::ROW;
                u.row = std::move(merged);
                out.items.push_back(std::move(u));
            }
        } else {
            auto m = marker_merger.merge({marker_versions.begin(), marker_versions.end()});
            if (m) {
                Unfiltered u;
                u.kind = Unfiltered::MARKER;
                u.marker = *m;
                out.items.push_back(std::move(u));
            }
        }
    }
    return out;
}

// ---------------------------------------------------------------------------
// purge (PurgeFunction.java:26-145, BTreeRow.purge:457-470, AbstractCell.purge:78-99)
// ---------------------------------------------------------------------------
struct Purger {
    int64_t now, gc_before;
    bool never_purge;
    const std::vector<PurgeRange>* overlaps;
    int64_t token;
    const bytes* key = nullptr;            // partition key (bloom probes)
    mutable int64_t kb = 0, ki = 0;        // murmur3_128 of key (h1, h0)
    mutable bool khash = false;

    // purgeEvaluator(key).test(ts) (CompactionController.java:247-286):
    // token range + (when the entry carries the sstable's bloom) a per-key
    // might-contain check, matching overlapIterator + BF.isPresent
    bool evaluator(int64_t ts) const {
        int64_t min_ts = INT64_MAX;
        bool has = false;
        for (const auto& r : *overlaps) {
            if (token < r.tok_lo || token > r.tok_hi) continue;
            if (!r.bloom.empty() && key) {
                if (!khash) {
                    uint64_t h[2];
                    murmur3_128_cassandra(key->data(), key->size(), 0, h);
                    kb = (int64_t)h[1];
                    ki = (int64_t)h[0];
                    khash = true;
                }
                uint64_t max = (uint64_t)r.bloom.size() * 8;
                int64_t base = kb, inc = ki;
                bool present = true;
                for (int i = 0; i < r.bloom_k; i++) {
                    int64_t m = base % (int64_t)max;
                    uint64_t idx = (uint64_t)((m ^ (m >> 63)) - (m >> 63));
                    if (!(r.bloom[idx >> 3] & (1u << (idx & 7)))) { present = false; break; }
                    base += inc;
                }
                if (!present) continue;  // source cannot contain this key
            }
            has = true;
            min_ts = std::min(min_ts, r.min_ts);
        }
        return !has || ts < min_ts;
    }
    bool should_purge(int64_t ts, int64_t ldt) const {  // the DeletionPurger lambda (PurgeFunction.java:40-44)
        if (never_purge) return false;
        return ldt < gc_before && evaluator(ts);
    }
    bool should_purge(const DeletionTime& dt) const {
        return !dt.live() && should_purge(dt.mfda, ldt_to_long(dt.ldt));
    }
    bool should_purge_liveness(const LivenessInfo& l) const {  // DeletionPurger.shouldPurge(liveness, now)
        bool is_live = !l.empty() && !l.expired() && (!l.expiring() || now < l.let);
        return !is_live && should_purge(l.ts, l.let);
    }
};

static std::optional<Cell> purge_cell(const Cell& c, const Purger& pg) {
    // AbstractCell.purge (AbstractCell.java:78-99)
    if (!c.is_live(pg.now)) {
        if (pg.should_purge(c.ts, ldt_to_long(c.ldt))) return std::nullopt;
        if (c.expiring()) {
            // convert expired cell to tombstone: ldt = localDeletionTime - ttl, then purge again
            Cell t;
            t.ts = c.ts;
            t.ldt = ldt_to_u32(ldt_to_long(c.ldt) - c.ttl);
            t.ttl = NO_TTL;
            // value dropped (BufferCell.tombstone has empty value)
            return purge_cell(t, pg);
        }
    }
    return c;
}

static bool purge_row(Row& r, const Purger& pg, bool enforce_strict_liveness) {
    // BTreeRow.purge (BTreeRow.java:457-470); unconditional application is
    // output-equivalent to the hasDeletion(nowInSec) fast path.
    if (pg.should_purge_liveness(r.live)) r.live = LivenessInfo{};
    if (pg.should_purge(r.del)) r.del = DeletionTime{};
    if (enforce_strict_liveness && r.del.live() && r.live.empty()) return false;
    bool any = false;
    for (auto& oc : r.cells) {
        if (!oc) continue;
        oc = purge_cell(*oc, pg);
        if (oc) any = true;
    }
    for (auto& ocd : r.complex) {
        if (!ocd) continue;
        // ComplexColumnData.purge (ComplexColumnData.java:212-216): purge the
        // complex deletion, purge each cell, null the column when empty
        if (pg.should_purge(ocd->del)) ocd->del = DeletionTime{};
        std::vector<Cell> kept;
        for (const Cell& c : ocd->cells) {
            auto p2 = purge_cell(c, pg);
            if (p2) { p2->path = c.path; kept.push_back(std::move(*p2)); }
        }
        ocd->cells = std::move(kept);
        if (ocd->del.live() && ocd->cells.empty()) ocd.reset();
        else any = true;
    }
    return !(r.live.empty() && r.del.live() && !any);  // empty row -> null
}

// ---------------------------------------------------------------------------
// GarbageSkippingUnfilteredRowIterator (CompactionIterator.java:401-598):
// remove data shadowed by the tombstone-source partition. Result merged with
// tombSource == merge of dataSource and tombSource.
// ---------------------------------------------------------------------------
static DeletionTime dt_max(const DeletionTime& a, const DeletionTime& b) {
    return a.supersedes(b) ? a : b;  // Ordering.natural over (mfda, ldt)
}
// BTreeRow.filter(ColumnFilter.all, activeDeletion, setActiveDeletionToRow=false)
static bool row_filter_active(Row& r, const DeletionTime& active) {
    bool may_shadow = !active.live() && !r.del.supersedes(active);
    if (may_shadow) {
        if (!r.live.empty() && r.live.ts <= active.mfda) r.live = LivenessInfo{};
        r.del = DeletionTime{};  // shadowed row deletion is dropped
        for (auto& c : r.cells)
            if (c && c->ts <= active.mfda) c.reset();
        // ComplexColumnData.filter(all, activeDeletion, null, -) — shadowed
        // complexDeletion dropped, cells dropped by activeDeletion.deletes
        // (ComplexColumnData.java:188-210); empty result -> column absent
        for (auto& cd : r.complex) {
            if (!cd) continue;
            if (active.supersedes(cd->del)) cd->del = DeletionTime{};
            std::vector<Cell> kept;
            for (auto& c : cd->cells)
                if (!(c.ts <= active.mfda)) kept.push_back(c);
            cd->cells = std::move(kept);
            if (cd->cells.empty() && cd->del.live()) cd.reset();
        }
    }
    return !row_is_empty(r);
}
// Cells.addNonShadowed identity decision for counter cells: the data cell is
// dropped iff Cells.reconcile(existing=a, update=b) returns b itself
// (Cells.java:178-189 with resolveCounter 121-162). ctx_merge's superset
// early-returns hand back the exact input bytes, so byte equality with an
// input reproduces the reference's object-identity check (left tested first).
static bool counter_b_wins(const Cell& a, const Cell& b) {
    bool at = a.ldt != LDT_NONE_U32, bt = b.ldt != LDT_NONE_U32;
    if (at || bt) {
        if (at != bt) return bt;  // tombstone beats any live counter
        return &reconcile_cells(a, b) == &b;  // two tombstones: regular rules
    }
    bool ae = a.value.empty(), be = b.value.empty();
    if (ae || be) {
        if (ae != be) return ae;  // non-empty wins (divergence note above)
        return !(a.ts > b.ts);
    }
    bytes m = ctx_merge(a.value, b.value);
    bool lid = m == a.value, rid = m == b.value;
    return rid && !lid && std::max(a.ts, b.ts) == b.ts;
}
// garbageFilterRow (CompactionIterator.java:544-556): row- or cell-level
static bool garbage_filter_row(Row& data, const Row& tomb, const DeletionTime& active,
                               bool cell_level, bool counters) {
    if (!cell_level) {
        return row_filter_active(data, dt_max(tomb.del, active));
    }
    // Rows.removeShadowedCells (Rows.java:266-316), simple columns
    DeletionTime deletion = dt_max(tomb.del, active);
    if (!data.live.empty() && data.live.ts <= deletion.mfda) data.live = LivenessInfo{};
    if (deletion.supersedes(data.del)) data.del = DeletionTime{};
    for (size_t i = 0; i < data.cells.size(); i++) {
        auto& a = data.cells[i];
        if (!a) continue;
        if (a->ts <= deletion.mfda) { a.reset(); continue; }
        const Cell* b = i < tomb.cells.size() && tomb.cells[i] ? &*tomb.cells[i] : nullptr;
        if (b) {
            bool bw = counters ? counter_b_wins(*a, *b) : &reconcile_cells(*a, *b) == b;
            if (bw) a.reset();  // overwritten by source
        }
    }
    // complex branch of removeShadowedCells (Rows.java:298-316): the data
    // complexDeletion is kept iff it supersedes max(updateDt, deletion) and
    // then raises the bar; per-path Cells.addNonShadowedComplex
    for (size_t i = 0; i < data.complex.size(); i++) {
        auto& a = data.complex[i];
        if (!a) continue;
        const ComplexData* b =
            i < tomb.complex.size() && tomb.complex[i] ? &*tomb.complex[i] : nullptr;
        DeletionTime updateDt = b ? b->del : DeletionTime{};
        DeletionTime maxDt = updateDt.supersedes(deletion) ? updateDt : deletion;
        DeletionTime keepDel{};  // LIVE unless data complexDeletion survives
        if (a->del.supersedes(maxDt)) { keepDel = a->del; maxDt = a->del; }
        std::vector<Cell> kept;
        size_t bi = 0;
        for (auto& c : a->cells) {
            const Cell* bc = nullptr;
            if (b) {
                while (bi < b->cells.size() && compare_cell_path(b->cells[bi].path, c.path) < 0)
                    bi++;
                if (bi < b->cells.size() && b->cells[bi].path == c.path) bc = &b->cells[bi];
            }
            if (c.ts <= maxDt.mfda) continue;  // deletion.deletes(existing)
            if (bc && &reconcile_cells(c, *bc) == bc) continue;
            kept.push_back(c);
        }
        if (kept.empty() && keepDel.live()) { a.reset(); continue; }
        a->del = keepDel;
        a->cells = std::move(kept);
    }
    return !row_is_empty(data);
}
static BoundKind close_bound_kind(BoundKind k) {
    return k == EXCL_END_INCL_START ? EXCL_END : k == INCL_END_EXCL_START ? INCL_END : k;
}
static BoundKind open_bound_kind(BoundKind k) {
    return k == EXCL_END_INCL_START ? INCL_START : k == INCL_END_EXCL_START ? EXCL_START : k;
}
static BoundKind invert_bound_kind(BoundKind k) {
    switch (k) {
        case INCL_END: return EXCL_START;
        case EXCL_END: return INCL_START;
        case INCL_START: return EXCL_END;
        case EXCL_START: return INCL_END;
        default: return k;
    }
}
static DeletionTime update_open_dt(const Unfiltered& u) {
    return u.marker.open(false) ? u.marker.open_dt() : DeletionTime{};
}

void garbage_filter(Partition& data, const Partition& tomb, const Header& h, bool cell_level) {
    bool counters = !h.regular_cols.empty() && is_counter_type(h.regular_cols[0].second);
    DeletionTime partition_dt = tomb.del;
    DeletionTime active = tomb.del;
    DeletionTime tomb_open, data_open, open_dt;  // LIVE
    if (!data.del.supersedes(tomb.del)) data.del = DeletionTime{};
    if (h.has_static() && !row_is_empty(data.static_row))
        if (!garbage_filter_row(data.static_row, tomb.static_row, active, cell_level, counters))
            data.static_row = Row{};
    std::vector<Unfiltered> out;
    size_t di = 0, ti = 0;
    auto emit_bound = [&](BoundKind k, const Clustering& vals, const DeletionTime& dt) {
        Unfiltered u;
        u.kind = Unfiltered::MARKER;
        u.marker.kind = k;
        u.marker.values = vals;
        u.marker.end_dt = dt;
        return u;
    };
    while (di < data.items.size()) {
        Unfiltered* next = nullptr;
        Unfiltered produced;
        bool have = false;
        int cmp = ti >= tomb.items.size()
                      ? -1
                      : compare_unfiltered(h, data.items[di], tomb.items[ti]);
        auto process_data_marker = [&]() -> bool {  // returns have
            data_open = update_open_dt(data.items[di]);
            bool before = open_dt.live();
            bool after = !data_open.supersedes(active);
            const Marker& m = data.items[di].marker;
            if (!before && !after) { produced = data.items[di]; return true; }
            if (!before && after) {
                produced = emit_bound(close_bound_kind(m.kind), m.values, m.close_dt());
                return true;
            }
            if (before && !after) {
                produced = emit_bound(open_bound_kind(m.kind), m.values, m.open_dt());
                return true;
            }
            return false;
        };
        if (cmp < 0) {
            if (data.items[di].kind == Unfiltered::ROW) {
                Row r = data.items[di].row;
                if (row_filter_active(r, active)) {
                    produced.kind = Unfiltered::ROW;
                    produced.row = std::move(r);
                    have = true;
                }
            } else {
                have = process_data_marker();
            }
        } else if (cmp == 0) {
            if (data.items[di].kind == Unfiltered::ROW) {
                Row r = data.items[di].row;
                if (garbage_filter_row(r, tomb.items[ti].row, active, cell_level, counters)) {
                    produced.kind = Unfiltered::ROW;
                    produced.row = std::move(r);
                    have = true;
                }
            } else {
                tomb_open = update_open_dt(tomb.items[ti]);
                active = dt_max(partition_dt, tomb_open);
                have = process_data_marker();
            }
        } else {
            if (tomb.items[ti].kind == Unfiltered::MARKER) {
                tomb_open = update_open_dt(tomb.items[ti]);
                active = dt_max(partition_dt, tomb_open);
                bool before = open_dt.live();
                bool after = !data_open.supersedes(active);
                // a suppressing deletion ends inside an open data range:
                // reopen the data range at the (inverted) tomb close bound
                if (before && !after) {
                    const Marker& tm = tomb.items[ti].marker;
                    produced = emit_bound(invert_bound_kind(close_bound_kind(tm.kind)),
                                          tm.values, data_open);
                    have = true;
                }
            }
        }
        if (have && produced.kind == Unfiltered::MARKER)
            open_dt = update_open_dt(produced);
        if (cmp <= 0) di++;
        if (cmp >= 0) ti++;
        if (have) out.push_back(std::move(produced));
        (void)next;
    }
    data.items = std::move(out);
}

bool purge_partition(Partition& p, int64_t now_sec, int64_t gc_before, bool never_purge,
                     const std::vector<PurgeRange>& overlaps, bool enforce_strict_liveness) {
    Purger pg{now_sec, gc_before, never_purge, &overlaps, p.token, &p.key};
    if (pg.should_purge(p.del)) p.del = DT_LIVE;
    if (!row_is_empty(p.static_row) && !purge_row(p.static_row, pg, false))
        p.static_row = Row{};
    std::vector<Unfiltered> kept;
    for (auto& u : p.items) {
        if (u.kind == Unfiltered::ROW) {
            if (purge_row(u.row, pg, enforce_strict_liveness)) kept.push_back(std::move(u));
        } else {
            // PurgeFunction.applyToMarker (PurgeFunction.java:113-144), reversed=false
            Marker& m = u.marker;
            if (m.boundary()) {
                bool purge_close = pg.should_purge(m.close_dt());
                bool purge_open = pg.should_purge(m.open_dt());
                if (purge_close && purge_open) continue;
                if (purge_close) {
                    // boundary -> corresponding open marker
                    Marker o;
                    o.kind = m.kind == EXCL_END_INCL_START ? INCL_START : EXCL_START;
                    o.values = m.values;
                    o.end_dt = m.start_dt;
                    u.marker = o;
                } else if (purge_open) {
                    Marker c;
                    c.kind = m.kind == EXCL_END_INCL_START ? EXCL_END : INCL_END;
                    c.values = m.values;
                    c.end_dt = m.end_dt;
                    u.marker = c;
                }
                kept.push_back(std::move(u));
            } else {
                if (!pg.should_purge(m.end_dt)) kept.push_back(std::move(u));
            }
        }
    }
    p.items = std::move(kept);
    return !(p.del.live() && p.items.empty() && row_is_empty(p.static_row));
}

// ---------------------------------------------------------------------------
// SerializationHeader.make (SerializationHeader.java:77-106)
// ---------------------------------------------------------------------------
static Header make_output_header(const std::vector<SSTable>& inputs) {
    std::vector<const SSTable*> ordered;
    for (auto& t : inputs) ordered.push_back(&t);
    std::stable_sort(ordered.begin(), ordered.end(),
                     [](const SSTable* a, const SSTable* b) { return a->generation > b->generation; });
    // EncodingStats.Collector over StatsMetadata mins, descending generation
    bool ts_set = false, ldt_set = false, ttl_set = false;
    int64_t min_ts = INT64_MAX, min_ldt = INT64_MAX;
    int32_t min_ttl = INT32_MAX;
    Header h;
    bool first = true;
    for (const SSTable* t : ordered) {
        ts_set = true; min_ts = std::min(min_ts, t->stats.min_timestamp);
        ldt_set = true; min_ldt = std::min(min_ldt, t->stats.min_ldt);
        ttl_set = true; min_ttl = std::min(min_ttl, t->stats.min_ttl);
        if (first) {
            h.key_type = t->header.key_type;
            h.clustering_types = t->header.clustering_types;
            first = false;
        }
        // columns.addAll — union preserving sorted-by-name Columns order
        for (auto cols_sel : {0, 1}) {
            auto& dst = cols_sel ? h.regular_cols : h.static_cols;
            auto& src = cols_sel ? t->header.regular_cols : t->header.static_cols;
            for (auto& c : src) {
                bool found = false;
                for (auto& d : dst) if (d.first == c.first) { found = true; break; }
                if (!found) dst.push_back(c);
            }
        }
    }
    std::sort(h.regular_cols.begin(), h.regular_cols.end(),
              [](auto& a, auto& b) { return a.first < b.first; });
    std::sort(h.static_cols.begin(), h.static_cols.end(),
              [](auto& a, auto& b) { return a.first < b.first; });
    // EncodingStats ctor epoch mapping (EncodingStats.java:78-89)
    h.stats.min_ts = (!ts_set || min_ts == NO_TIMESTAMP) ? TIMESTAMP_EPOCH : min_ts;
    h.stats.min_ldt = (!ldt_set || min_ldt == NO_EXPIRATION_TIME) ? DELETION_TIME_EPOCH : min_ldt;
    h.stats.min_ttl = !ttl_set ? 0 : min_ttl;
    return h;
}

// ---------------------------------------------------------------------------
// top-level compaction (CompactionTask.runMayThrow hot loop semantics)
// ---------------------------------------------------------------------------
CompactionResult compact(const CompactionJob& job) {
    CompactionResult res;
    res.out.header = make_output_header(job.inputs);
    res.out.comp = job.inputs.empty() ? CompressionParams{} : job.inputs[0].comp;
    res.merged_partition_counts.assign(job.inputs.size(), 0);

    size_t k = job.inputs.size();
    std::vector<size_t> pos(k, 0);
    while (true) {
        int min_src = -1;
        for (size_t i = 0; i < k; i++) {
            // shard / anticompaction-range filter: skip non-kept partitions
            while (pos[i] < job.inputs[i].parts.size() &&
                   (job.has_shard || !job.keep_ranges.empty())) {
                int64_t tok = job.inputs[i].parts[pos[i]].token;
                bool drop = job.has_shard && (tok < job.shard_lo || tok > job.shard_hi);
                if (!drop && !job.keep_ranges.empty()) {
                    bool in = false;
                    for (auto& r : job.keep_ranges)
                        if (tok >= r.tok_lo && tok <= r.tok_hi) { in = true; break; }
                    drop = (in == job.invert_ranges);
                }
                if (drop) pos[i]++;
                else break;
            }
            if (pos[i] >= job.inputs[i].parts.size()) continue;
            const Partition& p = job.inputs[i].parts[pos[i]];
            if (min_src < 0) { min_src = (int)i; continue; }
            const Partition& m = job.inputs[min_src].parts[pos[min_src]];
            if (compare_decorated_key(p.token, p.key.data(), p.key.size(),
                                      m.token, m.key.data(), m.key.size()) < 0)
                min_src = (int)i;
        }
        if (min_src < 0) break;
        const Partition& first = job.inputs[min_src].parts[pos[min_src]];
        std::vector<const Partition*> versions;
        for (size_t i = 0; i < k; i++) {
            if (pos[i] >= job.inputs[i].parts.size()) continue;
            const Partition& p = job.inputs[i].parts[pos[i]];
            if (compare_decorated_key(p.token, p.key.data(), p.key.size(),
                                      first.token, first.key.data(), first.key.size()) == 0) {
                versions.push_back(&p);
                pos[i]++;
            }
        }
        res.partitions_in += versions.size();
        res.merged_partition_counts[versions.size() - 1]++;
        for (auto* v : versions)
            for (auto& u : v->items) if (u.kind == Unfiltered::ROW) res.rows_in++;

        Partition merged = merge_partition_versions(versions, res.out.header);
        if (!job.tomb_sources.empty()) {
            // GarbageSkipper.applyToPartition: merge the shadow sources for
            // this key, then filter the merged data partition against them
            std::vector<const Partition*> tvers;
            for (auto& ts2 : job.tomb_sources) {
                auto it = std::lower_bound(
                    ts2.parts.begin(), ts2.parts.end(), merged,
                    [](const Partition& a, const Partition& b) {
                        return compare_decorated_key(a.token, a.key.data(), a.key.size(),
                                                     b.token, b.key.data(), b.key.size()) < 0;
                    });
                if (it != ts2.parts.end() && it->key == merged.key) tvers.push_back(&*it);
            }
            if (!tvers.empty()) {
                Partition tombm = merge_partition_versions(tvers, res.out.header);
                garbage_filter(merged, tombm, res.out.header, job.cell_level_gc);
            }
        }
        if (purge_partition(merged, job.now_sec, job.gc_before, job.never_purge,
                            job.overlaps, job.enforce_strict_liveness)) {
            res.partitions_out++;
            for (auto& u : merged.items) if (u.kind == Unfiltered::ROW) res.rows_out++;
            res.out.parts.push_back(std::move(merged));
        }
    }
    return res;
}

}  // namespace oracle
