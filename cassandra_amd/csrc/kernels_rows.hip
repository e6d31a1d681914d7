// PRODUCT — general (wide-partition) kernels: rows + range-tombstone markers
// under one fixed-width clustering column (or none). Replaces the round-1
// single-row kernels; the simple schema is the row_count<=1 special case of
// the same code path.
//
// Reference semantics restated here:
//   partition frame / row wire format: SortedTablePartitionWriter.java:97-166,
//     UnfilteredSerializer.java:36-305, Cell.java:240-306
//   merge: UnfilteredRowIterators.merge (:400-599), Row.Merger (Row.java:694-791),
//     Cells.reconcile (Cells.java:68-119), RangeTombstoneMarker.Merger
//     (RangeTombstoneMarker.java:72-198)
//   purge: PurgeFunction.java:26-145, CompactionController.java:247-286
//   promoted index: BigFormatPartitionWriter.java:128-245, IndexInfo.java:90-118,
//     RowIndexEntry.java:460-483,625-647
#pragma once
#include <hip/hip_runtime.h>
#include "codec.h"
#include "gpu_structs.h"

namespace gpuc {

__device__ inline void tomb_push(OutStats* st, uint32_t* ldts, uint32_t cap, uint32_t ldt) {
    unsigned long long i = atomicAdd(&st->tomb_count, 1ull);
    if (i < cap) ldts[i] = ldt;
}

// schema/runtime constants shared by the general kernels
struct SchemaParams {
    uint32_t n_ck;           // clustering columns (0..32)
    const int32_t* ck_w;     // per clustering column: 4/8 fixed or -1 variable
    uint32_t n_cols;         // SIMPLE regular columns (1..63; header order)
    const int32_t* col_fixed;  // per simple column: -1 variable else fixed width
    uint32_t n_cpx;          // 0 or 1: one complex column AFTER the simple ones
                             // (subset bitmap covers n_cols + n_cpx bits)
    uint32_t n_static;       // static columns (0 = schema has no statics)
    const int32_t* static_fixed;
    uint32_t column_index_size;  // promoted-index granularity (64 KiB default)
    // counter tables (every regular column CounterColumnType): reconcile
    // merges CounterContexts instead of timestamp resolution; merged
    // contexts are fresh bytes bump-allocated from ctr_arena (capacity =
    // total input counter-value bytes; st->cpx_alloc doubles as the bump —
    // counter schemas exclude complex columns)
    uint32_t counters;
    uint8_t* ctr_arena;
};

// sortable ck encoding: big-endian fixed-width signed value -> flip sign bit
__device__ inline uint64_t ck_sortable(const uint8_t* p, int width) {
    uint64_t v = 0;
    for (int b = 0; b < width; b++) v = (v << 8) | p[b];
    // values compare as signed integers of `width` bytes: flip the sign bit
    return v ^ (1ULL << (8 * width - 1));
}
// unsigned-lex sortable prefix of a variable-width clustering value
__device__ inline uint64_t ck_prefix_var(const uint8_t* p, uint32_t len) {
    uint64_t v = 0;
    uint32_t n = len < 8 ? len : 8;
    for (uint32_t b = 0; b < n; b++) v |= (uint64_t)p[b] << (8 * (7 - b));
    return v;
}
__device__ inline void ck_bytes(uint64_t ck, int width, uint8_t* out) {
    uint64_t v = ck ^ (1ULL << (8 * width - 1));
    for (int b = 0; b < width; b++) out[b] = (uint8_t)(v >> (8 * (width - 1 - b)));
}
// Kind.comparedToClustering (ClusteringPrefix.java): placement of a
// SHORTER prefix (bound) against longer prefixes sharing its components
__device__ inline int bk_cmp_to_clustering(uint8_t kk) {
    const int tbl[8] = {-1, -1, -1, -1, 0, 1, 1, 1};
    return tbl[kk];
}
// one clustering component (typed): fixed widths compare via the sortable
// u64; variable widths compare prefix then byte-walk (unsigned lex,
// shorter-first)
__device__ inline int ck_cmp_component(const UnfCols& u, uint64_t ac, uint64_t bc, int32_t w) {
    uint64_t pa = u.ck[ac], pb = u.ck[bc];
    if (pa != pb) return pa < pb ? -1 : 1;
    if (w >= 0) return 0;
    uint32_t la = u.ck_len[ac], lb = u.ck_len[bc];
    if (la > 8 || lb > 8) {
        const uint8_t* a8 = (const uint8_t*)u.ck_addr[ac];
        const uint8_t* b8 = (const uint8_t*)u.ck_addr[bc];
        uint32_t n = la < lb ? la : lb;
        for (uint32_t i = 8; i < n; i++)
            if (a8[i] != b8[i]) return a8[i] < b8[i] ? -1 : 1;
    }
    return la == lb ? 0 : (la < lb ? -1 : 1);
}
// ClusteringComparator over full prefixes (compare_clustering_prefix in the
// oracle; ClusteringComparator.compare in the reference): component-wise,
// then size, then Kind.comparison
__device__ inline int pos_cmp(const UnfCols& in, uint64_t a, uint64_t b, const SchemaParams& sch) {
    if (sch.n_ck) {
        uint32_t na = in.ck_count[a], nb = in.ck_count[b];
        uint32_t mn = na < nb ? na : nb;
        for (uint32_t c = 0; c < mn; c++) {
            int r = ck_cmp_component(in, a * sch.n_ck + c, b * sch.n_ck + c, sch.ck_w[c]);
            if (r) return r;
        }
        if (na != nb)
            return na < nb ? bk_cmp_to_clustering(in.rkind[a]) : -bk_cmp_to_clustering(in.rkind[b]);
    }
    return bk_comparison(in.rkind[a]) - bk_comparison(in.rkind[b]);
}

// ---------------------------------------------------------------------------
// parse pass A: one thread per partition — key/token/partition deletion,
// count unfiltereds (skip-walk via the size vints)
// ---------------------------------------------------------------------------
struct SrcDesc2 {
    const uint8_t* data;
    const uint64_t* part_pos;  // n_parts+1
    const uint64_t* part_end;  // scrub: explicit ends (else part_pos[li+1])
    uint32_t n_parts;
    int64_t min_ts, min_ldt;
    int32_t min_ttl;
    uint32_t rec_base;         // partition index base in the concatenated arrays
};

__global__ void k_parse_count(const SrcDesc2* srcs, uint32_t n_srcs, uint32_t total,
                              MRec* recs, ParsedCols pc, SchemaParams sp,
                              unsigned long long* error) {
    uint32_t gi = blockIdx.x * blockDim.x + threadIdx.x;
    if (gi >= total) return;
    uint32_t s = 0;
    while (s + 1 < n_srcs && gi >= srcs[s + 1].rec_base) s++;
    const SrcDesc2& sd = srcs[s];
    uint32_t li = gi - sd.rec_base;
    const uint8_t* base = sd.data;
    uint64_t pos = sd.part_pos[li];
    uint64_t end = sd.part_end ? sd.part_end[li] : sd.part_pos[li + 1];

    uint32_t klen = ((uint32_t)base[pos] << 8) | base[pos + 1];
    pos += 2;
    if (klen == 0) { atomicExch(error, 10ull); return; }
    uint64_t pfx = 0;
    uint32_t pb = klen < 8 ? klen : 8;
    for (uint32_t b = 0; b < pb; b++) pfx |= (uint64_t)base[pos + b] << (8 * (7 - b));
    int64_t token = murmur3_token(base + pos, klen);
    pc.key_addr[gi] = (uint64_t)(base + pos);
    pos += klen;
    MRec r{(uint64_t)token ^ 0x8000000000000000ULL, pfx, li, (uint16_t)s, (uint16_t)klen};
    recs[gi] = r;

    int64_t pdm = INT64_MIN;
    uint32_t pdl = LDT_NONE_U32;
    {
        uint8_t f = base[pos];
        if (f & 0x80) { pos++; if (f != 0x80) { atomicExch(error, 11ull); return; } }
        else {
            uint64_t v = 0;
            for (int i = 0; i < 8; i++) v = (v << 8) | base[pos + i];
            pdm = (int64_t)v;
            pdl = ((uint32_t)base[pos + 8] << 24) | ((uint32_t)base[pos + 9] << 16) |
                  ((uint32_t)base[pos + 10] << 8) | base[pos + 11];
            pos += 12;
        }
    }
    pc.pdel_mfda[gi] = pdm;
    pc.pdel_ldt[gi] = pdl;

    // static row slot (always present on disk when the schema has statics)
    if (sp.n_static) {
        uint8_t f = base[pos++];
        if (!(f & 0x80) || base[pos] != 0x01) { atomicExch(error, 18ull); return; }
        pos++;
        uint64_t size = uvint_get(base, &pos);
        uint64_t prev = uvint_get(base, &pos);
        pos += size - uvint_size(prev);
    }

    // skip-walk the unfiltereds
    uint32_t count = 0;
    while (true) {
        uint8_t flags = base[pos++];
        if (flags & 0x01) break;  // END_OF_PARTITION
        if ((flags & 0x80) || ((flags & 0x40) && !sp.n_cpx)) { atomicExch(error, 12ull); return; }  // extension / unexpected complex
        if (flags & 0x02) {
            // marker: kind, u16 size, values
            uint8_t kind = base[pos++];
            uint32_t nv = ((uint32_t)base[pos] << 8) | base[pos + 1];
            pos += 2;
            if (kind == BK_STATIC || nv > sp.n_ck || (nv && sp.n_ck == 0)) { atomicExch(error, 15ull); return; }
            if (nv == 0 && sp.n_ck != 0) { atomicExch(error, 16ull); return; }  // 0-value bound unsupported
            if (nv) {
                uint64_t hdr = uvint_get(base, &pos);
                if (hdr) { atomicExch(error, 17ull); return; }
                for (uint32_t c = 0; c < nv; c++)
                    pos += sp.ck_w[c] > 0 ? (uint64_t)sp.ck_w[c] : uvint_get(base, &pos);
            }
        } else if (sp.n_ck) {
            uint64_t hdr = uvint_get(base, &pos);
            if (hdr) { atomicExch(error, 17ull); return; }  // null/empty clustering unsupported
            for (uint32_t c = 0; c < sp.n_ck; c++)
                pos += sp.ck_w[c] > 0 ? (uint64_t)sp.ck_w[c] : uvint_get(base, &pos);
        }
        uint64_t size = uvint_get(base, &pos);
        uint64_t prev = uvint_get(base, &pos);
        pos += size - uvint_size(prev);  // body
        count++;
        if (pos > end) { atomicExch(error, 14ull); return; }
    }
    if (pos != end) { atomicExch(error, 14ull); return; }
    pc.row_count[gi] = count;
}

// ---------------------------------------------------------------------------
// parse pass B: full decode into the UnfCols arena (row_base prescanned)
// ---------------------------------------------------------------------------
// cpx_base: per-partition bases into the complex-cell arena (exscan of
// pc.cpx_total). nullptr = COUNT mode: walk everything, fill all non-arena
// outputs, and write pc.cpx_total[gi] for the sizing exscan.
__global__ void k_parse_rows(const SrcDesc2* srcs, uint32_t n_srcs, uint32_t total,
                             ParsedCols pc, UnfCols rc, SchemaParams sp,
                             unsigned long long* error, unsigned long long* rows_in,
                             const uint64_t* cpx_base = nullptr) {
    uint32_t gi = blockIdx.x * blockDim.x + threadIdx.x;
    if (gi >= total) return;
    uint32_t s = 0;
    while (s + 1 < n_srcs && gi >= srcs[s + 1].rec_base) s++;
    const SrcDesc2& sd = srcs[s];
    uint32_t li = gi - sd.rec_base;
    const uint8_t* base = sd.data;
    uint64_t pos = sd.part_pos[li];
    // skip key + partition deletion
    uint32_t klen = ((uint32_t)base[pos] << 8) | base[pos + 1];
    pos += 2 + klen;
    pos += (base[pos] & 0x80) ? 1 : 12;

    if (sp.n_static) {
        uint8_t flags = base[pos++];
        pos++;  // extended flags byte (IS_STATIC, validated in pass A)
        uvint_get(base, &pos);  // size
        uvint_get(base, &pos);  // prev
        uint8_t pf = 0;
        int64_t lts = NO_TIMESTAMP, llet = NO_DELETION_TIME, rdm = INT64_MIN;
        int32_t lttl = 0;
        uint32_t rdl = LDT_NONE_U32;
        if (flags & 0x04) { pf |= PF_LIVE_TS; lts = (int64_t)uvint_get(base, &pos) + sd.min_ts; }
        if (flags & 0x08) {
            lttl = (int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ttl;
            llet = (int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt;
        }
        if (flags & 0x10) {
            pf |= PF_ROW_DEL;
            rdm = (int64_t)uvint_get(base, &pos) + sd.min_ts;
            rdl = ldt_u32((int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt);
        }
        uint64_t missing = 0;
        if (!(flags & 0x20)) missing = uvint_get(base, &pos);
        bool any_cell = false;
        for (uint32_t c = 0; c < sp.n_static; c++) {
            uint64_t oc = (uint64_t)gi * sp.n_static + c;
            if (missing & (1ULL << c)) { pc.st.cell_flags[oc] = 0; continue; }
            any_cell = true;
            uint8_t cfl = CELLF_PRESENT;
            uint8_t cf = base[pos++];
            int64_t cts = (cf & 8) ? lts : (int64_t)uvint_get(base, &pos) + sd.min_ts;
            bool dead = cf & 1, exp = cf & 2;
            int64_t ldtl;
            if (cf & 16) ldtl = llet;
            else if (dead || exp) ldtl = (int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt;
            else ldtl = NO_DELETION_TIME;
            int32_t cttl = (cf & 16) ? lttl : (exp ? (int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ttl : 0);
            if (exp) cfl |= CELLF_EXPIRING;
            uint64_t vaddr = 0;
            uint32_t vlen = 0;
            if (!(cf & 4)) {
                cfl |= CELLF_HAS_VALUE;
                int32_t fw = sp.static_fixed[c];
                vlen = fw >= 0 ? (uint32_t)fw : (uint32_t)uvint_get(base, &pos);
                vaddr = (uint64_t)(base + pos);
                pos += vlen;
            }
            pc.st.cell_flags[oc] = cfl;
            pc.st.cell_ts[oc] = cts;
            pc.st.cell_ldt[oc] = ldt_u32(ldtl);
            pc.st.cell_ttl[oc] = cttl;
            pc.st.val_addr[oc] = vaddr;
            pc.st.val_len[oc] = vlen;
        }
        if ((pf & (PF_LIVE_TS | PF_ROW_DEL)) || any_cell) pf |= PF_HAS_ROW;
        pc.st.flags[gi] = pf;
        pc.st.live_ts[gi] = lts;
        pc.st.live_ttl[gi] = lttl;
        pc.st.live_let[gi] = llet;
        pc.st.rdel_mfda[gi] = rdm;
        pc.st.rdel_ldt[gi] = rdl;
    }

    uint64_t out = pc.row_base[gi];
    uint32_t emitted = 0;
    uint64_t cpx_cur = (sp.n_cpx && cpx_base) ? cpx_base[gi] : 0;
    uint32_t cpx_seen = 0;
    unsigned long long rows_local = 0;
    while (true) {
        uint8_t flags = base[pos++];
        if (flags & 0x01) break;
        uint64_t o = out + emitted;
        if (flags & 0x02) {
            // ---- marker ----
            uint8_t kind = base[pos++];
            uint32_t nv = ((uint32_t)base[pos] << 8) | base[pos + 1];
            pos += 2;
            if (nv) uvint_get(base, &pos);  // 32-batch header (0: all non-null)
            for (uint32_t c = 0; c < nv; c++) {
                uint64_t oc = o * sp.n_ck + c;
                uint32_t cklen;
                if (sp.ck_w[c] > 0) {
                    rc.ck[oc] = ck_sortable(base + pos, sp.ck_w[c]);
                    cklen = (uint32_t)sp.ck_w[c];
                } else {
                    cklen = (uint32_t)uvint_get(base, &pos);
                    rc.ck[oc] = ck_prefix_var(base + pos, cklen);
                }
                rc.ck_addr[oc] = (uint64_t)(base + pos);
                rc.ck_len[oc] = cklen;
                pos += cklen;
            }
            uvint_get(base, &pos);  // size
            uvint_get(base, &pos);  // prev
            rc.rkind[o] = kind;
            rc.ck_count[o] = (uint8_t)nv;
            rc.flags[o] = 0;
            // deltas: boundary = end then start; bound = single deletion
            int64_t em = (int64_t)uvint_get(base, &pos) + sd.min_ts;
            uint32_t el = ldt_u32((int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt);
            rc.rdel_mfda[o] = em;
            rc.rdel_ldt[o] = el;
            if (bk_is_boundary(kind)) {
                int64_t sm = (int64_t)uvint_get(base, &pos) + sd.min_ts;
                uint32_t sl = ldt_u32((int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt);
                rc.start_mfda[o] = sm;
                rc.start_ldt[o] = sl;
            } else {
                rc.start_mfda[o] = INT64_MIN;
                rc.start_ldt[o] = LDT_NONE_U32;
            }
            rc.live_ts[o] = NO_TIMESTAMP;
            rc.live_ttl[o] = 0;
            rc.live_let[o] = NO_DELETION_TIME;
            for (uint32_t c = 0; c < sp.n_cols; c++) rc.cell_flags[o * sp.n_cols + c] = 0;
            if (sp.n_cpx) {
                rc.cpx_del_mfda[o] = INT64_MIN;
                rc.cpx_del_ldt[o] = LDT_NONE_U32;
                rc.cpx_start[o] = cpx_cur;
                rc.cpx_count[o] = 0;
            }
        } else {
            // ---- row ----
            if (sp.n_ck) uvint_get(base, &pos);  // 32-batch header
            for (uint32_t c = 0; c < sp.n_ck; c++) {
                uint64_t oc = o * sp.n_ck + c;
                uint32_t cklen;
                if (sp.ck_w[c] > 0) {
                    rc.ck[oc] = ck_sortable(base + pos, sp.ck_w[c]);
                    cklen = (uint32_t)sp.ck_w[c];
                } else {
                    cklen = (uint32_t)uvint_get(base, &pos);
                    rc.ck[oc] = ck_prefix_var(base + pos, cklen);
                }
                rc.ck_addr[oc] = (uint64_t)(base + pos);
                rc.ck_len[oc] = cklen;
                pos += cklen;
            }
            uvint_get(base, &pos);  // size
            uvint_get(base, &pos);  // prev
            uint8_t pf = PF_HAS_ROW;
            int64_t lts = NO_TIMESTAMP, llet = NO_DELETION_TIME, rdm = INT64_MIN;
            int32_t lttl = 0;
            uint32_t rdl = LDT_NONE_U32;
            if (flags & 0x04) { pf |= PF_LIVE_TS; lts = (int64_t)uvint_get(base, &pos) + sd.min_ts; }
            if (flags & 0x08) {
                lttl = (int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ttl;
                llet = (int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt;
            }
            if (flags & 0x10) {
                pf |= PF_ROW_DEL;
                rdm = (int64_t)uvint_get(base, &pos) + sd.min_ts;
                rdl = ldt_u32((int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt);
            }
            // columns subset (Columns.serializeSubset: vint bitmap of MISSING)
            uint64_t missing = 0;
            if (!(flags & 0x20)) missing = uvint_get(base, &pos);
            for (uint32_t c = 0; c < sp.n_cols; c++) {
                uint64_t oc = o * sp.n_cols + c;
                if (missing & (1ULL << c)) {
                    rc.cell_flags[oc] = 0;
                    continue;
                }
                uint8_t cfl = CELLF_PRESENT;
                uint8_t cf = base[pos++];
                int64_t cts = (cf & 8) ? lts : (int64_t)uvint_get(base, &pos) + sd.min_ts;
                bool dead = cf & 1, exp = cf & 2;
                int64_t ldtl;
                if (cf & 16) ldtl = llet;
                else if (dead || exp) ldtl = (int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt;
                else ldtl = NO_DELETION_TIME;
                int32_t cttl = (cf & 16) ? lttl : (exp ? (int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ttl : 0);
                if (exp) cfl |= CELLF_EXPIRING;
                uint64_t vaddr = 0;
                uint32_t vlen = 0;
                if (!(cf & 4)) {
                    cfl |= CELLF_HAS_VALUE;
                    int32_t fw = sp.col_fixed[c];
                    vlen = fw >= 0 ? (uint32_t)fw : (uint32_t)uvint_get(base, &pos);
                    vaddr = (uint64_t)(base + pos);
                    pos += vlen;
                }
                rc.cell_flags[oc] = cfl;
                rc.cell_ts[oc] = cts;
                rc.cell_ldt[oc] = ldt_u32(ldtl);
                rc.cell_ttl[oc] = cttl;
                rc.val_addr[oc] = vaddr;
                rc.val_len[oc] = vlen;
            }
            // ---- complex column (bit sp.n_cols of the subset bitmap) ----
            if (sp.n_cpx) {
                rc.cpx_del_mfda[o] = INT64_MIN;
                rc.cpx_del_ldt[o] = LDT_NONE_U32;
                rc.cpx_count[o] = 0;
                rc.cpx_start[o] = cpx_cur;
                if (!(missing & (1ULL << sp.n_cols))) {
                    pf |= PF_HAS_CPX;
                    if (flags & 0x40) {  // HAS_COMPLEX_DELETION: per-column deletion
                        rc.cpx_del_mfda[o] = (int64_t)uvint_get(base, &pos) + sd.min_ts;
                        rc.cpx_del_ldt[o] = ldt_u32((int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt);
                    }
                    uint32_t ncc = (uint32_t)uvint_get(base, &pos);
                    rc.cpx_count[o] = ncc;
                    for (uint32_t e = 0; e < ncc; e++) {
                        uint8_t cf = base[pos++];
                        int64_t cts = (cf & 8) ? lts : (int64_t)uvint_get(base, &pos) + sd.min_ts;
                        bool dead = cf & 1, exp = cf & 2;
                        int64_t ldtl;
                        if (cf & 16) ldtl = llet;
                        else if (dead || exp) ldtl = (int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt;
                        else ldtl = NO_DELETION_TIME;
                        int32_t cttl = (cf & 16) ? lttl : (exp ? (int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ttl : 0);
                        uint32_t plen = (uint32_t)uvint_get(base, &pos);  // CellPath vint+bytes
                        uint64_t paddr = (uint64_t)(base + pos);
                        pos += plen;
                        uint64_t vaddr = 0;
                        uint32_t vlen = 0;
                        if (!(cf & 4)) {  // value: BytesType (map value), vint length
                            vlen = (uint32_t)uvint_get(base, &pos);
                            vaddr = (uint64_t)(base + pos);
                            pos += vlen;
                        }
                        if (cpx_base) {
                            uint64_t xe = cpx_cur + e;
                            rc.cpx.ts[xe] = cts;
                            rc.cpx.ldt[xe] = ldt_u32(ldtl);
                            rc.cpx.ttl[xe] = cttl;
                            rc.cpx.flags[xe] = CELLF_PRESENT | ((cf & 4) ? 0 : CELLF_HAS_VALUE) |
                                               (exp ? CELLF_EXPIRING : 0);
                            rc.cpx.path_addr[xe] = paddr;
                            rc.cpx.path_len[xe] = plen;
                            rc.cpx.val_addr[xe] = vaddr;
                            rc.cpx.val_len[xe] = vlen;
                        }
                    }
                    cpx_cur += ncc;
                    cpx_seen += ncc;
                }
            }
            rc.rkind[o] = BK_CLUSTERING;
            rc.ck_count[o] = (uint8_t)sp.n_ck;
            rc.flags[o] = pf;
            rc.live_ts[o] = lts;
            rc.live_ttl[o] = lttl;
            rc.live_let[o] = llet;
            rc.rdel_mfda[o] = rdm;
            rc.rdel_ldt[o] = rdl;
            rc.start_mfda[o] = INT64_MIN;
            rc.start_ldt[o] = LDT_NONE_U32;
            rows_local++;
        }
        emitted++;
    }
    if (rows_local) atomicAdd(rows_in, rows_local);  // one atomic per partition
    if (sp.n_cpx && pc.cpx_total) pc.cpx_total[gi] = cpx_seen;
    (void)error;
}

// MurmurHash.hash2_64 (utils/MurmurHash.java:96-150), the COMPACTION-HLL
// key hash (MetadataCollector.java:180-183); tail bytes sign-extended
// exactly as the reference's java bytes are. Oracle mirror:
// murmur2_64_cassandra (fixture-pinned).
__device__ inline uint64_t murmur2_64(const uint8_t* key, uint32_t length) {
    const uint64_t m = 0xc6a4a7935bd1e995ULL;
    const int r = 47;
    uint64_t h = m * (uint64_t)length;
    uint32_t nl = length >> 3;
    for (uint32_t i = 0; i < nl; i++) {
        uint64_t k;
        memcpy(&k, key + i * 8, 8);
        k *= m;
        k ^= k >> r;
        k *= m;
        h ^= k;
        h *= m;
    }
    uint32_t rem = length & 7;
    if (rem) {
        const uint8_t* t = key + length - rem;
        for (uint32_t b = rem; b-- > 0;)
            h ^= (uint64_t)(int64_t)(int8_t)t[b] << (8 * b);
        h *= m;
    }
    h ^= h >> r;
    h *= m;
    h ^= h >> r;
    return h;
}

__global__ void k_key_hash2(OutParts op, uint64_t n, uint64_t* out) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n) return;
    out[g] = op.keep[g] ? murmur2_64((const uint8_t*)op.key_addr[g], op.klen[g]) : 0;
}

// block-reduced merged-arity histogram (fed from op.merged_k)
__global__ void k_merged_hist(const uint8_t* merged_k, uint64_t n, OutStats* st) {
    __shared__ unsigned int sh[64];
    for (int i = threadIdx.x; i < 64; i += blockDim.x) sh[i] = 0;
    __syncthreads();
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g < n && merged_k[g]) atomicAdd(&sh[merged_k[g] - 1], 1u);
    __syncthreads();
    for (int i = threadIdx.x; i < 64; i += blockDim.x)
        if (sh[i]) atomicAdd(&st->merged_counts[i], (unsigned long long)sh[i]);
}

// ---------------------------------------------------------------------------
// repair-validation digests (Validator.rowHash + UnfilteredRowIterators/
// Rows/Cells/DeletionTime/ClusteringPrefix .digest): per kept output
// partition, concat(murmur3_128(1000), murmur3_128(2000)) over the field
// stream — big-endian ints/longs, inverted booleans, counter contexts
// body-only, cell localDeletionTime excluded. Oracle mirror:
// validator_digest (oracle/src/compact.cpp).
// ---------------------------------------------------------------------------
struct ValidateParams {
    const uint8_t* names;      // packed regular-column then static-column names
    const uint32_t* name_off;  // n_reg + n_static + 1 offsets
    uint32_t n_reg, n_static;
    uint32_t counters, n_cpx;
    uint32_t n_ck;
    const int32_t* ck_w;
};

__global__ void k_validate_digest(OutParts op, UnfCols out, uint64_t n, ValidateParams vp,
                                  uint8_t* hashes) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n || !op.keep[g]) return;
    M3Stream a, b;
    a.init(1000);
    b.init(2000);
    auto put = [&](const uint8_t* d, uint64_t m) { a.put(d, m); b.put(d, m); };
    auto u8 = [&](uint8_t v) { a.put_u8(v); b.put_u8(v); };
    auto i32 = [&](int32_t v) { a.put_i32be(v); b.put_i32be(v); };
    auto i64 = [&](int64_t v) { a.put_i64be(v); b.put_i64be(v); };
    auto bl = [&](bool v) { a.put_bool(v); b.put_bool(v); };
    auto put_ck_comp = [&](uint64_t oc) {
        if (out.ck_addr[oc]) {
            put((const uint8_t*)out.ck_addr[oc], out.ck_len[oc]);
        } else {
            int w = vp.ck_w[oc % (vp.n_ck ? vp.n_ck : 1)];
            uint64_t v = out.ck[oc] ^ (1ull << (8 * w - 1));
            uint8_t bts[8];
            for (int i = 0; i < w; i++) bts[i] = (uint8_t)(v >> (8 * (w - 1 - i)));
            put(bts, w);
        }
    };
    auto put_cell_fields = [&](int64_t ts, int32_t ttl, const uint8_t* val, uint32_t vlen,
                               bool counter_cell, const uint8_t* path, uint32_t plen) {
        if (counter_cell) {
            if (vlen >= 2) {
                int16_t hn = (int16_t)((uint16_t)((uint16_t)val[0] << 8) | val[1]);
                uint32_t hl = 2 + (uint32_t)(hn < 0 ? -hn : hn) * 2;
                if (vlen > hl) put(val + hl, vlen - hl);
            }
        } else {
            put(val, vlen);
        }
        i64(ts);
        i32(ttl);
        bl(counter_cell);
        if (path) put(path, plen);
    };
    // partition key + deletion + column names + reverse flag
    put((const uint8_t*)op.key_addr[g], op.klen[g]);
    i64(op.pdel_mfda[g]);
    for (uint32_t c = 0; c < vp.n_reg; c++)
        put(vp.names + vp.name_off[c], vp.name_off[c + 1] - vp.name_off[c]);
    bool static_present = vp.n_static && (op.st.flags[g] & PF_HAS_ROW);
    if (static_present)
        for (uint32_t c = vp.n_reg; c < vp.n_reg + vp.n_static; c++)
            put(vp.names + vp.name_off[c], vp.name_off[c + 1] - vp.name_off[c]);
    bl(false);  // isReverseOrder
    // static row (EMPTY_STATIC_ROW when absent)
    {
        u8(0);
        u8(3);  // STATIC kind
        if (static_present) {
            i64(op.st.rdel_mfda[g]);
            bl(false);
            i64((op.st.flags[g] & PF_LIVE_TS) ? op.st.live_ts[g] : NO_TIMESTAMP);
            for (uint32_t c = 0; c < vp.n_static; c++) {
                uint64_t oc = g * vp.n_static + c;
                if (!(op.st.cell_flags[oc] & CELLF_PRESENT)) continue;
                put_cell_fields(op.st.cell_ts[oc], op.st.cell_ttl[oc],
                                (const uint8_t*)op.st.val_addr[oc], op.st.val_len[oc], false,
                                nullptr, 0);
            }
        } else {
            i64(INT64_MIN);
            bl(false);
            i64(NO_TIMESTAMP);
        }
    }
    // unfiltereds
    uint64_t rb = op.row_base[g];
    for (uint32_t j = 0; j < op.row_count[g]; j++) {
        uint64_t o = rb + j;
        uint8_t kind = out.rkind[o];
        if (kind == BK_CLUSTERING) {
            u8(0);
            for (uint32_t c = 0; c < vp.n_ck; c++) put_ck_comp(o * vp.n_ck + c);
            u8(4);
            uint8_t f = out.flags[o];
            i64((f & PF_ROW_DEL) ? out.rdel_mfda[o] : INT64_MIN);
            bl(false);
            i64((f & PF_LIVE_TS) ? out.live_ts[o] : NO_TIMESTAMP);
            for (uint32_t c = 0; c < vp.n_reg - vp.n_cpx; c++) {
                uint64_t oc = o * (vp.n_reg - vp.n_cpx) + c;
                if (!(out.cell_flags[oc] & CELLF_PRESENT)) continue;
                put_cell_fields(out.cell_ts[oc], out.cell_ttl[oc],
                                (const uint8_t*)out.val_addr[oc], out.val_len[oc],
                                vp.counters && out.cell_ldt[oc] == LDT_NONE_U32, nullptr, 0);
            }
            if (vp.n_cpx && (f & PF_HAS_CPX)) {
                bool del_live = out.cpx_del_mfda[o] == INT64_MIN && out.cpx_del_ldt[o] == LDT_NONE_U32;
                if (!del_live) i64(out.cpx_del_mfda[o]);
                uint64_t si = out.cpx_start[o];
                for (uint32_t e = 0; e < out.cpx_count[o]; e++) {
                    uint64_t xe = si + e;
                    put_cell_fields(out.cpx.ts[xe], out.cpx.ttl[xe],
                                    (const uint8_t*)out.cpx.val_addr[xe], out.cpx.val_len[xe],
                                    false, (const uint8_t*)out.cpx.path_addr[xe],
                                    out.cpx.path_len[xe]);
                }
            }
        } else {
            u8(1);  // RANGE_TOMBSTONE_MARKER
            for (uint32_t c = 0; c < out.ck_count[o]; c++) put_ck_comp(o * vp.n_ck + c);
            u8(kind);
            i64(out.rdel_mfda[o]);
            if (bk_is_boundary(kind)) i64(out.start_mfda[o]);
        }
    }
    a.final16(hashes + g * 32);
    b.final16(hashes + g * 32 + 16);
}

// total present-cell value bytes (counter arena sizing)
__global__ void k_sum_vallen(UnfCols in, uint64_t n_cells, unsigned long long* out) {
    __shared__ unsigned long long sh;
    if (threadIdx.x == 0) sh = 0;
    __syncthreads();
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    unsigned long long v = 0;
    if (i < n_cells && (in.cell_flags[i] & CELLF_PRESENT)) v = in.val_len[i];
    if (v) atomicAdd(&sh, v);
    __syncthreads();
    if (threadIdx.x == 0 && sh) atomicAdd(out, sh);
}

__global__ void k_widen_u32(const uint32_t* in, uint64_t* out, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) out[i] = in[i];
}

// ---------------------------------------------------------------------------
// group input row totals (for the output arena layout)
// ---------------------------------------------------------------------------
__global__ void k_group_row_sums(const MRec* recs, const uint64_t* group_start,
                                 uint64_t n_groups, uint64_t n_recs,
                                 const uint32_t* src_bases, ParsedCols pc,
                                 uint64_t* group_rows) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    uint64_t beg = group_start[g];
    uint64_t endi = g + 1 < n_groups ? group_start[g + 1] : n_recs;
    uint64_t sum = 0;
    for (uint64_t m = beg; m < endi; m++)
        sum += pc.row_count[src_bases[recs[m].src] + recs[m].idx];
    group_rows[g] = sum;
}

// ---------------------------------------------------------------------------
// reconcile + purge, general: one thread per group merges the ≤k versions'
// row/marker streams and purges, writing merged unfiltereds to the out arena.
// ---------------------------------------------------------------------------
struct PurgeParams2 {
    int64_t now_sec, gc_before;
    int32_t never_purge;
    int32_t enforce_strict_liveness;
    const int64_t* ov_lo;
    const int64_t* ov_hi;
    const int64_t* ov_min_ts;
    // optional per-overlap bloom filters (packed u32 words + per-entry word
    // offset/bit-length/k; bit_len 0 = interval-only entry): enables the
    // per-key evaluator of CompactionController.getPurgeEvaluator. kb/ki
    // hold the CURRENT partition key's murmur3_128 (h1, h0), filled by the
    // kernel into its by-value copy before any should_purge2 call.
    const uint32_t* ov_bloom_words;
    const uint64_t* ov_bloom_off;
    const uint64_t* ov_bloom_bits;
    const int32_t* ov_bloom_k;
    int32_t ov_has_bloom;
    int64_t kb, ki;
    int32_t n_overlaps;
    int32_t has_shard;
    int64_t shard_lo, shard_hi;
    // multi-range keep filter (anticompaction: CompactionManager.antiCompactGroup
    // splits data by repaired ranges — one pass keeps tokens IN the ranges,
    // the inverted pass keeps the complement)
    const int64_t* kr_lo;
    const int64_t* kr_hi;
    int32_t n_keep_ranges;
    int32_t invert_ranges;
};

__device__ inline bool token_kept(const PurgeParams2& pp, int64_t token) {
    if (pp.has_shard && (token < pp.shard_lo || token > pp.shard_hi)) return false;
    if (pp.n_keep_ranges) {
        bool in = false;
        for (int i = 0; i < pp.n_keep_ranges && !in; i++)
            in = token >= pp.kr_lo[i] && token <= pp.kr_hi[i];
        if (in == (pp.invert_ranges != 0)) return false;
    }
    return true;
}

__device__ inline bool purge_eval2(const PurgeParams2& pp, int64_t token, int64_t ts) {
    int64_t min_ts = INT64_MAX;
    bool has = false;
    for (int i = 0; i < pp.n_overlaps; i++) {
        if (token < pp.ov_lo[i] || token > pp.ov_hi[i]) continue;
        if (pp.ov_has_bloom && pp.ov_bloom_bits[i]) {
            // BloomFilter.isPresent on the overlapping sstable: skip the
            // entry when it cannot contain this key
            const uint32_t* w = pp.ov_bloom_words + pp.ov_bloom_off[i];
            int64_t base = pp.kb, inc = pp.ki;
            bool present = true;
            for (int b = 0; b < pp.ov_bloom_k[i]; b++) {
                int64_t m = base % (int64_t)pp.ov_bloom_bits[i];
                uint64_t idx = (uint64_t)((m ^ (m >> 63)) - (m >> 63));
                if (!(w[idx >> 5] & (1u << (idx & 31)))) { present = false; break; }
                base += inc;
            }
            if (!present) continue;
        }
        has = true;
        min_ts = min(min_ts, pp.ov_min_ts[i]);
    }
    return !has || ts < min_ts;
}
__device__ inline bool should_purge2(const PurgeParams2& pp, int64_t token, int64_t ts, int64_t ldt) {
    if (pp.never_purge) return false;
    return ldt < pp.gc_before && purge_eval2(pp, token, ts);
}
__device__ inline bool dt_sup(int64_t am, uint32_t al, int64_t bm, uint32_t bl) {
    return am > bm || (am == bm && ldt_long(al) > ldt_long(bl));
}

// Cell.compareValues tie-break: unsigned lexicographic, shorter-first
__device__ inline int cmp_values(uint64_t la, uint32_t ll, uint64_t ra, uint32_t rl) {
    const uint8_t* lp = (const uint8_t*)la;
    const uint8_t* rp = (const uint8_t*)ra;
    uint32_t nn = ll < rl ? ll : rl;
    for (uint32_t x = 0; x < nn; x++)
        if (lp[x] != rp[x]) return lp[x] < rp[x] ? -1 : 1;
    return ll == rl ? 0 : (ll < rl ? -1 : 1);
}

// ---------------------------------------------------------------------------
// CounterContext device machinery (db/context/CounterContext.java): walker
// over (header flags, 16B-id shards) + the k-way merge equivalent of the
// reference's pairwise chain (equal by commutativity/associativity of the
// compare() lattice for live clocks; remote clocks are never 0 in practice)
// ---------------------------------------------------------------------------
struct CtxSt {
    const uint8_t* p;
    uint32_t len, body, off, hoff;
    bool neg_hdr, g, l;
    __device__ void update() {
        g = l = false;
        if (hoff < body) {
            int16_t e = (int16_t)((uint16_t)((uint16_t)p[hoff] << 8) | p[hoff + 1]);
            int32_t idx = (int32_t)((off - body) / 32);
            if (!neg_hdr && e == (int16_t)(idx + INT16_MIN)) g = true;
            else if (e == (int16_t)idx) l = true;
        }
    }
    __device__ void init(const uint8_t* q, uint32_t n) {
        p = q;
        len = n;
        int16_t h = n >= 2 ? (int16_t)((uint16_t)((uint16_t)q[0] << 8) | q[1]) : 0;
        neg_hdr = h < 0;
        uint32_t hn = (uint32_t)(h < 0 ? -h : h);
        body = 2 + hn * 2;
        off = body;
        hoff = 2;
        update();
    }
    __device__ bool has() const { return off + 32 <= len; }
    __device__ void next() {
        if (g || l) hoff += 2;
        off += 32;
        update();
    }
    __device__ const uint8_t* id() const { return p + off; }
    __device__ int64_t clock() const {
        uint64_t v = 0;
        for (int b = 0; b < 8; b++) v = (v << 8) | p[off + 16 + b];
        return (int64_t)v;
    }
    __device__ int64_t count() const {
        uint64_t v = 0;
        for (int b = 0; b < 8; b++) v = (v << 8) | p[off + 24 + b];
        return (int64_t)v;
    }
};
__device__ inline int ctx_idcmp(const uint8_t* a, const uint8_t* b) {
    for (int i = 0; i < 16; i++)
        if (a[i] != b[i]) return a[i] < b[i] ? -1 : 1;
    return 0;
}
// remote shard "left wins" rule (CounterContext.compare, both-remote tail)
__device__ inline bool ctx_remote_better(int64_t lc, int64_t lk, int64_t rc, int64_t rk) {
    if (lc == rc) return lk > rk;
    return (lc >= 0 && rc > 0 && lc >= rc) || (lc < 0 && (rc > 0 || lc < rc));
}

#define GPUC_MAX_ARITY 64

// MA = max merge arity this instantiation supports (member-stream arrays are
// MA-sized locals; small MA keeps them out of scratch — the host picks the
// smallest instantiation >= n_inputs)
template <int MA>
__global__ void k_reconcile_rows(const MRec* recs, const uint64_t* group_start,
                                 uint64_t n_groups, uint64_t n_recs,
                                 const uint32_t* src_bases, ParsedCols pc, UnfCols in,
                                 OutParts op, UnfCols out, const uint64_t* out_base,
                                 SchemaParams sp, PurgeParams2 pp, OutStats* st,
                                 unsigned long long* error) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    uint64_t beg = group_start[g];
    uint64_t endi = g + 1 < n_groups ? group_start[g + 1] : n_recs;
    uint32_t k = (uint32_t)(endi - beg);
    // merged-arity histogram: a direct atomicAdd here serializes millions of
    // threads on one address (~150 ms at C2's 14.9M single-row groups);
    // store the arity and let k_merged_hist block-reduce it instead
    op.merged_k[g] = (uint8_t)(k > 64 ? 64 : k);
    if (k > (uint32_t)MA) { atomicExch(error, 20ull); return; }

    const MRec r0 = recs[beg];
    int64_t token = (int64_t)(r0.tok ^ 0x8000000000000000ULL);
    op.keypfx[g] = r0.pfx;
    op.key_addr[g] = pc.key_addr[src_bases[r0.src] + r0.idx];
    op.token[g] = token;
    op.klen[g] = r0.klen;
    op.row_base[g] = out_base[g];
    op.row_count[g] = 0;
    if (!token_kept(pp, token)) { op.keep[g] = 0; return; }
    if (pp.ov_has_bloom) {
        uint64_t h[2];
        murmur3_128((const uint8_t*)op.key_addr[g], r0.klen, 0, h);
        pp.kb = (int64_t)h[1];
        pp.ki = (int64_t)h[0];
    }

    // member streams
    uint64_t mb[MA];
    uint32_t mcnt[MA], mpos[MA];
    int64_t pdm = INT64_MIN;
    uint32_t pdl = LDT_NONE_U32;
    #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
        uint32_t i = src_bases[recs[beg + m].src] + recs[beg + m].idx;
        mb[m] = pc.row_base[i];
        mcnt[m] = pc.row_count[i];
        mpos[m] = 0;
        if (!dt_sup(pdm, pdl, pc.pdel_mfda[i], pc.pdel_ldt[i])) { pdm = pc.pdel_mfda[i]; pdl = pc.pdel_ldt[i]; }
    }
    bool pdel_live0 = pdm == INT64_MIN && pdl == LDT_NONE_U32;
    if (!pdel_live0 && should_purge2(pp, token, pdm, ldt_long(pdl))) { pdm = INT64_MIN; pdl = LDT_NONE_U32; }
    // NOTE: purge of the partition deletion happens AFTER merge in the
    // reference pipeline, but merge uses the UNPURGED value as activeDeletion.
    int64_t adm = INT64_MIN;   // merged partition deletion for shadow filtering
    uint32_t adl = LDT_NONE_U32;
    {
        // recompute un-purged merged value (pdm/pdl may have been purged above)
        int64_t m2 = INT64_MIN;
        uint32_t l2 = LDT_NONE_U32;
        #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
            uint32_t i = src_bases[recs[beg + m].src] + recs[beg + m].idx;
            if (!dt_sup(m2, l2, pc.pdel_mfda[i], pc.pdel_ldt[i])) { m2 = pc.pdel_mfda[i]; l2 = pc.pdel_ldt[i]; }
        }
        adm = m2;
        adl = l2;
    }

    // ---- static row merge (Row.Merger over the versions' static rows with
    // activeDeletion = the UNPURGED merged partition deletion) + purge ----
    bool static_kept = false;
    if (sp.n_static) {
        const uint32_t NS = sp.n_static;
        uint8_t of = 0;
        int64_t lts = NO_TIMESTAMP, llet = NO_DELETION_TIME, rdm = INT64_MIN;
        int32_t lttl = 0;
        uint32_t rdl = LDT_NONE_U32;
        uint64_t sslot = g;  // output partition index
        bool any_cell = false;
        // gather present versions
        uint32_t nvers = 0;
        uint32_t iver[GPUC_MAX_ARITY > 16 ? 64 : 64];
        #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
            uint32_t i = src_bases[recs[beg + m].src] + recs[beg + m].idx;
            if (pc.st.flags[i] & PF_HAS_ROW) iver[nvers++] = i;
        }
        // k==1: UnfilteredRowIterators.merge of one iterator passes the
        // static row through unfiltered (as the row path does); otherwise the
        // Row.Merger single-version shortcut needs a live activeDeletion
        if (nvers == 1 && (k == 1 || (adm == INT64_MIN && adl == LDT_NONE_U32))) {
            uint32_t i = iver[0];
            of = pc.st.flags[i];
            lts = pc.st.live_ts[i]; lttl = pc.st.live_ttl[i]; llet = pc.st.live_let[i];
            rdm = pc.st.rdel_mfda[i]; rdl = pc.st.rdel_ldt[i];
            for (uint32_t c = 0; c < NS; c++) {
                uint64_t dc = sslot * NS + c, sc = (uint64_t)i * NS + c;
                uint8_t cfl = pc.st.cell_flags[sc];
                op.st.cell_flags[dc] = cfl;
                if (!(cfl & CELLF_PRESENT)) continue;
                any_cell = true;
                op.st.cell_ts[dc] = pc.st.cell_ts[sc];
                op.st.cell_ldt[dc] = pc.st.cell_ldt[sc];
                op.st.cell_ttl[dc] = pc.st.cell_ttl[sc];
                op.st.val_addr[dc] = pc.st.val_addr[sc];
                op.st.val_len[dc] = pc.st.val_len[sc];
            }
        } else if (nvers > 0) {
            bool has_live = false;
            for (uint32_t v = 0; v < nvers; v++) {
                uint32_t i = iver[v];
                uint8_t f = pc.st.flags[i];
                if (f & PF_LIVE_TS) {
                    int64_t t2 = pc.st.live_ts[i];
                    int32_t ttl2 = pc.st.live_ttl[i];
                    int64_t let2 = pc.st.live_let[i];
                    bool sup;
                    if (!has_live) sup = true;
                    else if (t2 != lts) sup = t2 > lts;
                    else {
                        bool e1 = lttl == INT32_MAX, e2 = ttl2 == INT32_MAX;
                        if (e1 != e2) sup = e2;
                        else if ((lttl != 0) == (ttl2 != 0)) sup = let2 > llet;
                        else sup = ttl2 != 0;
                    }
                    if (sup) { lts = t2; lttl = ttl2; llet = let2; has_live = true; }
                }
                if (f & PF_ROW_DEL)
                    if (dt_sup(pc.st.rdel_mfda[i], pc.st.rdel_ldt[i], rdm, rdl)) { rdm = pc.st.rdel_mfda[i]; rdl = pc.st.rdel_ldt[i]; }
            }
            if (has_live) of |= PF_LIVE_TS;
            int64_t am2 = adm;
            uint32_t al2 = adl;
            bool row_del_kept = false;
            if (dt_sup(rdm, rdl, am2, al2)) { am2 = rdm; al2 = rdl; row_del_kept = true; }
            if (!row_del_kept) { rdm = INT64_MIN; rdl = LDT_NONE_U32; }
            else of |= PF_ROW_DEL;
            if (has_live && lts <= am2) { of &= ~PF_LIVE_TS; lts = NO_TIMESTAMP; lttl = 0; llet = NO_DELETION_TIME; }
            for (uint32_t c = 0; c < NS; c++) {
                int64_t cts = NO_TIMESTAMP;
                uint32_t cldt = LDT_NONE_U32;
                int32_t cttl = 0;
                uint64_t va = 0;
                uint32_t vl = 0;
                bool have_cell = false, cell_val = false, cell_exp = false;
                for (uint32_t v = 0; v < nvers; v++) {
                    uint64_t o = (uint64_t)iver[v] * NS + c;
                    uint8_t f = pc.st.cell_flags[o];
                    if (!(f & CELLF_PRESENT)) continue;
                    int64_t ts2 = pc.st.cell_ts[o];
                    if (ts2 <= am2) continue;
                    if (!have_cell) {
                        have_cell = true;
                        cts = ts2; cldt = pc.st.cell_ldt[o]; cttl = pc.st.cell_ttl[o];
                        va = pc.st.val_addr[o]; vl = pc.st.val_len[o];
                        cell_val = f & CELLF_HAS_VALUE;
                        cell_exp = f & CELLF_EXPIRING;
                        continue;
                    }
                    bool take_right = false;
                    uint32_t rl = pc.st.cell_ldt[o];
                    bool l_dt = cldt != LDT_NONE_U32, r_dt = rl != LDT_NONE_U32;
                    if (cts != ts2) take_right = ts2 > cts;
                    else if (l_dt || r_dt) {
                        if (l_dt != r_dt) take_right = r_dt;
                        else {
                            bool l_tomb = !cell_exp, r_tomb = !(f & CELLF_EXPIRING);
                            if (l_tomb != r_tomb) take_right = r_tomb;
                            else if (cldt != rl) take_right = ldt_long(rl) > ldt_long(cldt);
                            else take_right = cmp_values(va, vl, pc.st.val_addr[o], pc.st.val_len[o]) < 0;
                        }
                    } else {
                        take_right = cmp_values(va, vl, pc.st.val_addr[o], pc.st.val_len[o]) < 0;
                    }
                    if (take_right) {
                        cts = ts2; cldt = rl; cttl = pc.st.cell_ttl[o];
                        va = pc.st.val_addr[o]; vl = pc.st.val_len[o];
                        cell_val = f & CELLF_HAS_VALUE;
                        cell_exp = f & CELLF_EXPIRING;
                    }
                }
                uint64_t dc = sslot * NS + c;
                if (have_cell) {
                    any_cell = true;
                    op.st.cell_flags[dc] = CELLF_PRESENT | (cell_val ? CELLF_HAS_VALUE : 0) |
                                           (cell_exp ? CELLF_EXPIRING : 0);
                    op.st.cell_ts[dc] = cts;
                    op.st.cell_ldt[dc] = cldt;
                    op.st.cell_ttl[dc] = cttl;
                    op.st.val_addr[dc] = va;
                    op.st.val_len[dc] = vl;
                } else {
                    op.st.cell_flags[dc] = 0;
                }
            }
            if ((of & (PF_LIVE_TS | PF_ROW_DEL)) || any_cell) of |= PF_HAS_ROW;
            else of = 0;
        } else {
            for (uint32_t c = 0; c < NS; c++) op.st.cell_flags[sslot * NS + c] = 0;
        }
        // purge (BTreeRow.purge over the merged static row)
        if (of & PF_HAS_ROW) {
            if (of & PF_LIVE_TS) {
                bool is_live = lttl == INT32_MAX ? false : (lttl != 0 ? pp.now_sec < llet : true);
                if (!is_live && should_purge2(pp, token, lts, llet)) { of &= ~PF_LIVE_TS; lts = NO_TIMESTAMP; lttl = 0; llet = NO_DELETION_TIME; }
            }
            if ((of & PF_ROW_DEL) && rdm != INT64_MIN && should_purge2(pp, token, rdm, ldt_long(rdl))) { of &= ~PF_ROW_DEL; rdm = INT64_MIN; rdl = LDT_NONE_U32; }
            any_cell = false;
            for (uint32_t c = 0; c < NS; c++) {
                uint64_t dc = sslot * NS + c;
                uint8_t cfl = op.st.cell_flags[dc];
                if (!(cfl & CELLF_PRESENT)) continue;
                int64_t cts = op.st.cell_ts[dc];
                uint32_t cldt = op.st.cell_ldt[dc];
                int32_t cttl = op.st.cell_ttl[dc];
                bool live_cell = cldt == LDT_NONE_U32 || (cttl != 0 && pp.now_sec < ldt_long(cldt));
                if (!live_cell) {
                    if (should_purge2(pp, token, cts, ldt_long(cldt))) { op.st.cell_flags[dc] = 0; continue; }
                    if (cttl != 0) {
                        int64_t nldt = ldt_long(cldt) - cttl;
                        if (should_purge2(pp, token, cts, nldt)) { op.st.cell_flags[dc] = 0; continue; }
                        op.st.cell_ldt[dc] = ldt_u32(nldt);
                        op.st.cell_ttl[dc] = 0;
                        op.st.cell_flags[dc] = CELLF_PRESENT;
                        op.st.val_len[dc] = 0;
                    }
                }
                any_cell = true;
            }
            if (!(of & (PF_LIVE_TS | PF_ROW_DEL)) && !any_cell) of = 0;
        }
        op.st.flags[g] = of;
        op.st.live_ts[g] = lts;
        op.st.live_ttl[g] = lttl;
        op.st.live_let[g] = llet;
        op.st.rdel_mfda[g] = rdm;
        op.st.rdel_ldt[g] = rdl;
        static_kept = (of & PF_HAS_ROW) != 0;
    }

    // marker-merger state (RangeTombstoneMarker.Merger)
    int64_t om_m[GPUC_MAX_ARITY];
    uint32_t om_l[GPUC_MAX_ARITY];
    uint64_t om_set = 0;  // bit per member

    auto merged_open = [&](int64_t* mm, uint32_t* ml) {
        // currentOpenDeletionTimeInMerged: biggest open not superseding pdel -> LIVE
        int64_t bm = INT64_MIN;
        uint32_t bl = LDT_NONE_U32;
        bool any = false;
        for (uint32_t m = 0; m < k; m++)
            if (om_set & (1ULL << m))
                if (!any || dt_sup(om_m[m], om_l[m], bm, bl)) { bm = om_m[m]; bl = om_l[m]; any = true; }
        if (!any || !dt_sup(bm, bl, adm, adl)) { *mm = INT64_MIN; *ml = LDT_NONE_U32; }
        else { *mm = bm; *ml = bl; }
    };

    uint64_t obase = out_base[g];
    uint32_t ocount = 0;
    uint64_t cur_fo = 0;  // row index whose clustering the next emit copies
    // cells_written: the row path fills out.cell_* at slot obase+ocount
    // BEFORE calling emit; markers pass false and get all-absent cells
    auto emit = [&](uint8_t kind, uint64_t ck, uint8_t flags, int64_t lts, int32_t lttl,
                    int64_t llet, int64_t rdm, uint32_t rdl, int64_t smf, uint32_t sld,
                    bool cells_written) {
        (void)ck;
        uint64_t o = obase + ocount++;
        if (!cells_written)
            for (uint32_t c = 0; c < sp.n_cols; c++) out.cell_flags[o * sp.n_cols + c] = 0;
        out.rkind[o] = kind;
        uint32_t nv = in.ck_count[cur_fo];
        out.ck_count[o] = (uint8_t)nv;
        for (uint32_t c = 0; c < nv; c++) {
            uint64_t dc = o * sp.n_ck + c, sc = cur_fo * sp.n_ck + c;
            out.ck[dc] = in.ck[sc];
            out.ck_addr[dc] = in.ck_addr[sc];
            out.ck_len[dc] = in.ck_len[sc];
        }
        out.flags[o] = flags;
        out.live_ts[o] = lts;
        out.live_ttl[o] = lttl;
        out.live_let[o] = llet;
        out.rdel_mfda[o] = rdm;
        out.rdel_ldt[o] = rdl;
        out.start_mfda[o] = smf;
        out.start_ldt[o] = sld;
    };

    // purge helpers
    auto purge_dt = [&](int64_t m, uint32_t l) { return m != INT64_MIN && should_purge2(pp, token, m, ldt_long(l)); };

    while (true) {
        // find min position; track the winner's VALUE so the MA-sized
        // stream arrays never see a runtime index (keeps them in registers)
        int first = -1;
        uint64_t fo = 0;
        #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
            if (mpos[m] >= mcnt[m]) continue;
            uint64_t o = mb[m] + mpos[m];
            if (first < 0 || pos_cmp(in, o, fo, sp) < 0) { first = (int)m; fo = o; }
        }
        if (first < 0) break;
        uint64_t fck = in.ck[fo * (sp.n_ck ? sp.n_ck : 1)];
        cur_fo = fo;
        uint8_t fkind = in.rkind[fo];
        bool is_row = bk_comparison(fkind) == 2;
        // gather members at this position
        uint64_t members = 0;
        uint32_t nmem = 0;
        int lastm = -1;
        uint64_t last_o = 0;
        #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
            if (mpos[m] >= mcnt[m]) continue;
            uint64_t o = mb[m] + mpos[m];
            if (pos_cmp(in, o, fo, sp) == 0) {
                members |= 1ULL << m;
                nmem++;
                lastm = (int)m;
                last_o = o;
            }
        }

        if (is_row) {
            // active deletion for rows = open marker in merged stream, else pdel
            int64_t am, al_;
            uint32_t al;
            {
                int64_t mm;
                uint32_t ml;
                merged_open(&mm, &ml);
                if (mm == INT64_MIN && ml == LDT_NONE_U32) { am = adm; al = adl; }
                else { am = mm; al = ml; }
                al_ = 0;
                (void)al_;
            }
            bool active_live = am == INT64_MIN && al == LDT_NONE_U32;
            // merge row versions (Row.Merger); k==1 group passthrough applies
            // at the PARTITION level (UnfilteredRowIterators.merge of 1 iterator)
            uint8_t of = 0;
            int64_t lts = NO_TIMESTAMP, llet = NO_DELETION_TIME, rdm = INT64_MIN;
            int32_t lttl = 0;
            uint32_t rdl = LDT_NONE_U32;
            const uint32_t NC = sp.n_cols;
            uint64_t oslot = obase + ocount;  // cells written here pre-emit
            bool any_cell = false;
            // merged complex column (ColumnDataReducer complex branch)
            int64_t xdm = INT64_MIN;
            uint32_t xdl = LDT_NONE_U32;
            uint64_t xstart = 0;
            uint32_t xcnt = 0;
            if (k == 1 || (nmem == 1 && active_live)) {
                uint64_t o = last_o;
                of = in.flags[o];
                lts = in.live_ts[o]; lttl = in.live_ttl[o]; llet = in.live_let[o];
                rdm = in.rdel_mfda[o]; rdl = in.rdel_ldt[o];
                for (uint32_t c = 0; c < NC; c++) {
                    uint64_t ic = o * NC + c, ocx = oslot * NC + c;
                    uint8_t cfl = in.cell_flags[ic];
                    out.cell_flags[ocx] = cfl;
                    if (!(cfl & CELLF_PRESENT)) continue;
                    any_cell = true;
                    out.cell_ts[ocx] = in.cell_ts[ic];
                    out.cell_ldt[ocx] = in.cell_ldt[ic];
                    out.cell_ttl[ocx] = in.cell_ttl[ic];
                    out.val_addr[ocx] = in.val_addr[ic];
                    out.val_len[ocx] = in.val_len[ic];
                }
                if (sp.n_cpx && (of & PF_HAS_CPX)) {
                    xdm = in.cpx_del_mfda[o];
                    xdl = in.cpx_del_ldt[o];
                    uint32_t nc2 = in.cpx_count[o];
                    xstart = atomicAdd(&st->cpx_alloc, (unsigned long long)nc2);
                    uint64_t si = in.cpx_start[o];
                    for (uint32_t e2 = 0; e2 < nc2; e2++) {
                        out.cpx.ts[xstart + e2] = in.cpx.ts[si + e2];
                        out.cpx.ldt[xstart + e2] = in.cpx.ldt[si + e2];
                        out.cpx.ttl[xstart + e2] = in.cpx.ttl[si + e2];
                        out.cpx.flags[xstart + e2] = in.cpx.flags[si + e2];
                        out.cpx.path_addr[xstart + e2] = in.cpx.path_addr[si + e2];
                        out.cpx.path_len[xstart + e2] = in.cpx.path_len[si + e2];
                        out.cpx.val_addr[xstart + e2] = in.cpx.val_addr[si + e2];
                        out.cpx.val_len[xstart + e2] = in.cpx.val_len[si + e2];
                    }
                    xcnt = nc2;
                }
            } else {
                bool has_live = false;
                #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
                    if (!(members & (1ULL << m))) continue;
                    uint64_t o = mb[m] + mpos[m];
                    uint8_t f = in.flags[o];
                    if (f & PF_LIVE_TS) {
                        int64_t t2 = in.live_ts[o];
                        int32_t ttl2 = in.live_ttl[o];
                        int64_t let2 = in.live_let[o];
                        bool sup;
                        if (!has_live) sup = true;
                        else if (t2 != lts) sup = t2 > lts;
                        else {
                            bool e1 = lttl == INT32_MAX, e2 = ttl2 == INT32_MAX;
                            if (e1 != e2) sup = e2;
                            else if ((lttl != 0) == (ttl2 != 0)) sup = let2 > llet;
                            else sup = ttl2 != 0;
                        }
                        if (sup) { lts = t2; lttl = ttl2; llet = let2; has_live = true; }
                    }
                    if (f & PF_ROW_DEL)
                        if (dt_sup(in.rdel_mfda[o], in.rdel_ldt[o], rdm, rdl)) { rdm = in.rdel_mfda[o]; rdl = in.rdel_ldt[o]; }
                }
                if (has_live) of |= PF_LIVE_TS;
                int64_t am2 = am;
                uint32_t al2 = al;
                bool row_del_kept = false;
                if (dt_sup(rdm, rdl, am2, al2)) { am2 = rdm; al2 = rdl; row_del_kept = true; }
                if (!row_del_kept) { rdm = INT64_MIN; rdl = LDT_NONE_U32; }
                else of |= PF_ROW_DEL;
                if (has_live && lts <= am2) { of &= ~PF_LIVE_TS; lts = NO_TIMESTAMP; lttl = 0; llet = NO_DELETION_TIME; }
                // per-column Cells.reconcile (Cells.java:145-179) + activeDeletion filter
                for (uint32_t c = 0; c < NC && sp.counters; c++) {
                    // Cells.resolveCounter: tombstones beat any live counter
                    // (then regular rules among tombstones); empty values
                    // lose; else k-way CounterContext merge into the arena
                    bool any_tomb = false, any_val = false;
                    uint32_t nv3 = 0;
                    uint64_t mem_oc[MA];
                    #pragma unroll
                    for (uint32_t m = 0; m < (uint32_t)MA; m++) {
                        if (m >= k) break;
                        if (!(members & (1ULL << m))) continue;
                        uint64_t o = (mb[m] + mpos[m]) * NC + c;
                        uint8_t f2 = in.cell_flags[o];
                        if (!(f2 & CELLF_PRESENT)) continue;
                        if (in.cell_ts[o] <= am2) continue;
                        if (in.cell_ldt[o] != LDT_NONE_U32) any_tomb = true;
                        else if (in.val_len[o] > 0) any_val = true;
                        mem_oc[nv3++] = o;
                    }
                    uint64_t ocx = oslot * NC + c;
                    if (nv3 == 0) { out.cell_flags[ocx] = 0; continue; }
                    any_cell = true;
                    if (any_tomb) {
                        // regular reconcile restricted to the tombstones
                        bool have = false;
                        int64_t cts = NO_TIMESTAMP;
                        uint32_t cldt = LDT_NONE_U32;
                        uint64_t va = 0;
                        uint32_t vl = 0;
                        for (uint32_t m2 = 0; m2 < nv3; m2++) {
                            uint64_t o = mem_oc[m2];
                            if (in.cell_ldt[o] == LDT_NONE_U32) continue;
                            int64_t ts2 = in.cell_ts[o];
                            uint32_t rl = in.cell_ldt[o];
                            bool take = !have ||
                                        (cts != ts2 ? ts2 > cts
                                                    : (cldt != rl ? ldt_long(rl) > ldt_long(cldt)
                                                                  : cmp_values(va, vl, in.val_addr[o], in.val_len[o]) < 0));
                            if (take) { have = true; cts = ts2; cldt = rl; va = in.val_addr[o]; vl = in.val_len[o]; }
                        }
                        out.cell_flags[ocx] = CELLF_PRESENT | (vl ? CELLF_HAS_VALUE : 0);
                        out.cell_ts[ocx] = cts;
                        out.cell_ldt[ocx] = cldt;
                        out.cell_ttl[ocx] = 0;
                        out.val_addr[ocx] = va;
                        out.val_len[ocx] = vl;
                        continue;
                    }
                    int64_t max_ts = NO_TIMESTAMP;
                    uint64_t first_nonempty = 0;
                    uint32_t n_live = 0, total_len = 0;
                    for (uint32_t m2 = 0; m2 < nv3; m2++) {
                        uint64_t o = mem_oc[m2];
                        if (in.cell_ts[o] > max_ts) max_ts = in.cell_ts[o];
                        if (in.val_len[o] > 0) {
                            if (!n_live) first_nonempty = o;
                            mem_oc[n_live++] = o;  // compact non-empty to front
                            total_len += in.val_len[o];
                        }
                    }
                    out.cell_flags[ocx] = CELLF_PRESENT;
                    out.cell_ts[ocx] = max_ts;
                    out.cell_ldt[ocx] = LDT_NONE_U32;
                    out.cell_ttl[ocx] = 0;
                    if (!any_val || n_live == 0) {
                        out.val_addr[ocx] = 0;
                        out.val_len[ocx] = 0;
                        continue;
                    }
                    if (n_live == 1) {
                        out.cell_flags[ocx] |= CELLF_HAS_VALUE;
                        out.val_addr[ocx] = in.val_addr[first_nonempty];
                        out.val_len[ocx] = in.val_len[first_nonempty];
                        continue;
                    }
                    // k-way context merge, two passes (count, then write)
                    uint8_t* dst = nullptr;
                    uint32_t wr = 0, nflag = 0, nsh = 0;
                    for (int pass = 0; pass < 2; pass++) {
                        CtxSt st2[MA];
                        for (uint32_t m2 = 0; m2 < n_live; m2++) {
                            uint64_t o = mem_oc[m2];
                            st2[m2].init((const uint8_t*)in.val_addr[o], in.val_len[o]);
                        }
                        if (pass == 1) {
                            uint64_t base = atomicAdd(&st->cpx_alloc, (unsigned long long)(2 + nflag * 2 + nsh * 32));
                            dst = sp.ctr_arena + base;
                            dst[0] = (uint8_t)(nflag >> 8);
                            dst[1] = (uint8_t)nflag;
                            wr = 2 + nflag * 2;
                            nflag = 0;
                            nsh = 0;
                        }
                        while (true) {
                            const uint8_t* minid = nullptr;
                            for (uint32_t m2 = 0; m2 < n_live; m2++)
                                if (st2[m2].has() && (!minid || ctx_idcmp(st2[m2].id(), minid) < 0))
                                    minid = st2[m2].id();
                            if (!minid) break;
                            bool hg = false, hl = false;
                            int64_t bc = 0, bk = 0, suml_c = 0, suml_k = 0, rc = 0, rk = 0;
                            bool hr = false;
                            const uint8_t* sid = minid;
                            for (uint32_t m2 = 0; m2 < n_live; m2++) {
                                if (!st2[m2].has() || ctx_idcmp(st2[m2].id(), minid) != 0) continue;
                                int64_t ck3 = st2[m2].clock(), cn3 = st2[m2].count();
                                if (st2[m2].g) {
                                    if (!hg || ck3 > bc || (ck3 == bc && cn3 > bk)) { bc = ck3; bk = cn3; }
                                    hg = true;
                                } else if (st2[m2].l) {
                                    hl = true;
                                    suml_c += ck3;
                                    suml_k += cn3;
                                } else {
                                    if (!hr || ctx_remote_better(ck3, cn3, rc, rk)) { rc = ck3; rk = cn3; }
                                    hr = true;
                                }
                                st2[m2].next();
                            }
                            int64_t oc3, ok3;
                            int role;  // 0 global 1 local 2 remote
                            if (hg) { oc3 = bc; ok3 = bk; role = 0; }
                            else if (hl) { oc3 = suml_c; ok3 = suml_k; role = 1; }
                            else { oc3 = rc; ok3 = rk; role = 2; }
                            if (pass == 0) {
                                if (role != 2) nflag++;
                                nsh++;
                            } else {
                                if (role != 2) {
                                    int16_t e = role == 0 ? (int16_t)((int32_t)nsh + INT16_MIN)
                                                          : (int16_t)nsh;
                                    dst[2 + nflag * 2] = (uint8_t)((uint16_t)e >> 8);
                                    dst[3 + nflag * 2] = (uint8_t)e;
                                    nflag++;
                                }
                                for (int b = 0; b < 16; b++) dst[wr + b] = sid[b];
                                for (int b = 7; b >= 0; b--) dst[wr + 16 + (7 - b)] = (uint8_t)((uint64_t)oc3 >> (8 * b));
                                for (int b = 7; b >= 0; b--) dst[wr + 24 + (7 - b)] = (uint8_t)((uint64_t)ok3 >> (8 * b));
                                wr += 32;
                                nsh++;
                            }
                        }
                    }
                    out.cell_flags[ocx] |= CELLF_HAS_VALUE;
                    out.val_addr[ocx] = (uint64_t)dst;
                    out.val_len[ocx] = wr;
                }
                for (uint32_t c = 0; c < NC && !sp.counters; c++) {
                    int64_t cts = NO_TIMESTAMP;
                    uint32_t cldt = LDT_NONE_U32;
                    int32_t cttl = 0;
                    uint64_t va = 0;
                    uint32_t vl = 0;
                    bool have_cell = false, cell_val = false, cell_exp = false;
                    #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
                        if (!(members & (1ULL << m))) continue;
                        uint64_t o = (mb[m] + mpos[m]) * NC + c;
                        uint8_t f = in.cell_flags[o];
                        if (!(f & CELLF_PRESENT)) continue;
                        int64_t ts2 = in.cell_ts[o];
                        if (ts2 <= am2) continue;
                        if (!have_cell) {
                            have_cell = true;
                            cts = ts2; cldt = in.cell_ldt[o]; cttl = in.cell_ttl[o];
                            va = in.val_addr[o]; vl = in.val_len[o];
                            cell_val = f & CELLF_HAS_VALUE;
                            cell_exp = f & CELLF_EXPIRING;
                            continue;
                        }
                        bool take_right = false;
                        uint32_t rl = in.cell_ldt[o];
                        bool l_dt = cldt != LDT_NONE_U32, r_dt = rl != LDT_NONE_U32;
                        if (cts != ts2) take_right = ts2 > cts;
                        else if (l_dt || r_dt) {
                            if (l_dt != r_dt) take_right = r_dt;
                            else {
                                bool l_tomb = !cell_exp, r_tomb = !(f & CELLF_EXPIRING);
                                if (l_tomb != r_tomb) take_right = r_tomb;
                                else if (cldt != rl) take_right = ldt_long(rl) > ldt_long(cldt);
                                else take_right = cmp_values(va, vl, in.val_addr[o], in.val_len[o]) < 0;
                            }
                        } else {
                            take_right = cmp_values(va, vl, in.val_addr[o], in.val_len[o]) < 0;
                        }
                        if (take_right) {
                            cts = ts2; cldt = rl; cttl = in.cell_ttl[o];
                            va = in.val_addr[o]; vl = in.val_len[o];
                            cell_val = f & CELLF_HAS_VALUE;
                            cell_exp = f & CELLF_EXPIRING;
                        }
                    }
                    uint64_t ocx = oslot * NC + c;
                    if (have_cell) {
                        any_cell = true;
                        out.cell_flags[ocx] = CELLF_PRESENT | (cell_val ? CELLF_HAS_VALUE : 0) |
                                              (cell_exp ? CELLF_EXPIRING : 0);
                        out.cell_ts[ocx] = cts;
                        out.cell_ldt[ocx] = cldt;
                        out.cell_ttl[ocx] = cttl;
                        out.val_addr[ocx] = va;
                        out.val_len[ocx] = vl;
                    } else {
                        out.cell_flags[ocx] = 0;
                    }
                }
                if (sp.n_cpx) {
                    // complexDeletion = max over versions; supersedes(active)
                    // decides both the cell filter and whether it is kept
                    // (Row.java:857-874)
                    bool anyv = false;
                    #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
                        if (!(members & (1ULL << m))) continue;
                        uint64_t o = mb[m] + mpos[m];
                        if (!(in.flags[o] & PF_HAS_CPX)) continue;
                        anyv = true;
                        if (dt_sup(in.cpx_del_mfda[o], in.cpx_del_ldt[o], xdm, xdl)) {
                            xdm = in.cpx_del_mfda[o];
                            xdl = in.cpx_del_ldt[o];
                        }
                    }
                    if (anyv) {
                        int64_t cam = am2;
                        uint32_t cal = al2;
                        bool keep_del = dt_sup(xdm, xdl, am2, al2);
                        if (keep_del) { cam = xdm; cal = xdl; }
                        else { xdm = INT64_MIN; xdl = LDT_NONE_U32; }
                        uint32_t ppos[MA], pcnt[MA];
                        uint64_t pbase[MA];
                        uint32_t nv2 = 0, totc = 0;
                        #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
                            if (!(members & (1ULL << m))) continue;
                            uint64_t o = mb[m] + mpos[m];
                            if (!(in.flags[o] & PF_HAS_CPX)) continue;
                            pbase[nv2] = in.cpx_start[o];
                            pcnt[nv2] = in.cpx_count[o];
                            ppos[nv2] = 0;
                            totc += in.cpx_count[o];
                            nv2++;
                        }
                        xstart = atomicAdd(&st->cpx_alloc, (unsigned long long)totc);
                        auto cmp_path = [&](uint64_t a, uint64_t b) -> int {
                            return cmp_values(in.cpx.path_addr[a], in.cpx.path_len[a],
                                              in.cpx.path_addr[b], in.cpx.path_len[b]);
                        };
                        while (true) {
                            int fi = -1;
                            uint64_t fxe = 0;
                            #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= nv2) break;
                                if (ppos[m] >= pcnt[m]) continue;
                                uint64_t xe0 = pbase[m] + ppos[m];
                                if (fi < 0 || cmp_path(xe0, fxe) < 0) { fi = (int)m; fxe = xe0; }
                            }
                            if (fi < 0) break;
                            // CellReducer over same-path cells in version order
                            bool have = false;
                            int64_t cts = NO_TIMESTAMP;
                            uint32_t cldt = LDT_NONE_U32;
                            int32_t cttl = 0;
                            uint64_t va = 0, pa = 0;
                            uint32_t vl = 0, pl = 0;
                            bool cell_val = false, cell_exp = false;
                            #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= nv2) break;
                                if (ppos[m] >= pcnt[m]) continue;
                                uint64_t xe = pbase[m] + ppos[m];
                                if (cmp_path(xe, fxe) != 0) continue;
                                ppos[m]++;
                                int64_t ts2 = in.cpx.ts[xe];
                                if (cam != INT64_MIN && ts2 <= cam) continue;  // activeDeletion.deletes
                                uint8_t f2 = in.cpx.flags[xe];
                                if (!have) {
                                    have = true;
                                    cts = ts2; cldt = in.cpx.ldt[xe]; cttl = in.cpx.ttl[xe];
                                    va = in.cpx.val_addr[xe]; vl = in.cpx.val_len[xe];
                                    pa = in.cpx.path_addr[xe]; pl = in.cpx.path_len[xe];
                                    cell_val = f2 & CELLF_HAS_VALUE;
                                    cell_exp = f2 & CELLF_EXPIRING;
                                    continue;
                                }
                                bool take_right = false;
                                uint32_t rl = in.cpx.ldt[xe];
                                bool l_dt = cldt != LDT_NONE_U32, r_dt = rl != LDT_NONE_U32;
                                if (cts != ts2) take_right = ts2 > cts;
                                else if (l_dt || r_dt) {
                                    if (l_dt != r_dt) take_right = r_dt;
                                    else {
                                        bool l_tomb = !cell_exp, r_tomb = !(f2 & CELLF_EXPIRING);
                                        if (l_tomb != r_tomb) take_right = r_tomb;
                                        else if (cldt != rl) take_right = ldt_long(rl) > ldt_long(cldt);
                                        else take_right = cmp_values(va, vl, in.cpx.val_addr[xe], in.cpx.val_len[xe]) < 0;
                                    }
                                } else {
                                    take_right = cmp_values(va, vl, in.cpx.val_addr[xe], in.cpx.val_len[xe]) < 0;
                                }
                                if (take_right) {
                                    cts = ts2; cldt = rl; cttl = in.cpx.ttl[xe];
                                    va = in.cpx.val_addr[xe]; vl = in.cpx.val_len[xe];
                                    cell_val = f2 & CELLF_HAS_VALUE;
                                    cell_exp = f2 & CELLF_EXPIRING;
                                }
                            }
                            if (have) {
                                uint64_t xe2 = xstart + xcnt++;
                                out.cpx.ts[xe2] = cts;
                                out.cpx.ldt[xe2] = cldt;
                                out.cpx.ttl[xe2] = cttl;
                                out.cpx.flags[xe2] = CELLF_PRESENT | (cell_val ? CELLF_HAS_VALUE : 0) |
                                                     (cell_exp ? CELLF_EXPIRING : 0);
                                out.cpx.path_addr[xe2] = pa;
                                out.cpx.path_len[xe2] = pl;
                                out.cpx.val_addr[xe2] = va;
                                out.cpx.val_len[xe2] = vl;
                            }
                        }
                        // Builder.build: live deletion + no cells -> null column
                        if (xcnt || xdm != INT64_MIN || xdl != LDT_NONE_U32) {
                            of |= PF_HAS_CPX;
                            any_cell = true;
                        }
                    }
                }
                if ((of & (PF_LIVE_TS | PF_ROW_DEL)) || any_cell) of |= PF_HAS_ROW;
                else of = 0;
            }
            // ---- purge the merged row (BTreeRow.purge + AbstractCell.purge) ----
            if (of & PF_HAS_ROW) {
                if (of & PF_LIVE_TS) {
                    bool is_live = lttl == INT32_MAX ? false : (lttl != 0 ? pp.now_sec < llet : true);
                    if (!is_live && should_purge2(pp, token, lts, llet)) { of &= ~PF_LIVE_TS; lts = NO_TIMESTAMP; lttl = 0; llet = NO_DELETION_TIME; }
                }
                if ((of & PF_ROW_DEL) && purge_dt(rdm, rdl)) { of &= ~PF_ROW_DEL; rdm = INT64_MIN; rdl = LDT_NONE_U32; }
                any_cell = false;
                for (uint32_t c = 0; c < NC; c++) {
                    uint64_t ocx = oslot * NC + c;
                    uint8_t cfl = out.cell_flags[ocx];
                    if (!(cfl & CELLF_PRESENT)) continue;
                    int64_t cts = out.cell_ts[ocx];
                    uint32_t cldt = out.cell_ldt[ocx];
                    int32_t cttl = out.cell_ttl[ocx];
                    bool live_cell = cldt == LDT_NONE_U32 || (cttl != 0 && pp.now_sec < ldt_long(cldt));
                    if (!live_cell) {
                        if (should_purge2(pp, token, cts, ldt_long(cldt))) { out.cell_flags[ocx] = 0; continue; }
                        if (cttl != 0) {
                            int64_t nldt = ldt_long(cldt) - cttl;
                            if (should_purge2(pp, token, cts, nldt)) { out.cell_flags[ocx] = 0; continue; }
                            out.cell_ldt[ocx] = ldt_u32(nldt);
                            out.cell_ttl[ocx] = 0;
                            out.cell_flags[ocx] = CELLF_PRESENT;  // expired -> tombstone, value dropped
                            out.val_len[ocx] = 0;
                        }
                    }
                    any_cell = true;
                }
                if (sp.n_cpx && (of & PF_HAS_CPX)) {
                    // ComplexColumnData.purge (ComplexColumnData.java:212-216)
                    if (xdm != INT64_MIN && should_purge2(pp, token, xdm, ldt_long(xdl))) {
                        xdm = INT64_MIN;
                        xdl = LDT_NONE_U32;
                    }
                    uint32_t w = 0;
                    for (uint32_t e2 = 0; e2 < xcnt; e2++) {
                        uint64_t xe = xstart + e2;
                        int64_t cts = out.cpx.ts[xe];
                        uint32_t cldt = out.cpx.ldt[xe];
                        int32_t cttl = out.cpx.ttl[xe];
                        uint8_t cfl = out.cpx.flags[xe];
                        uint32_t vl = out.cpx.val_len[xe];
                        bool live_cell = cldt == LDT_NONE_U32 || (cttl != 0 && pp.now_sec < ldt_long(cldt));
                        if (!live_cell) {
                            if (should_purge2(pp, token, cts, ldt_long(cldt))) continue;
                            if (cttl != 0) {
                                int64_t nldt = ldt_long(cldt) - cttl;
                                if (should_purge2(pp, token, cts, nldt)) continue;
                                cldt = ldt_u32(nldt);
                                cttl = 0;
                                cfl = CELLF_PRESENT;  // expired -> tombstone, value dropped
                                vl = 0;
                            }
                        }
                        uint64_t xo = xstart + w++;
                        out.cpx.ts[xo] = cts;
                        out.cpx.ldt[xo] = cldt;
                        out.cpx.ttl[xo] = cttl;
                        out.cpx.flags[xo] = cfl;
                        out.cpx.path_addr[xo] = out.cpx.path_addr[xe];
                        out.cpx.path_len[xo] = out.cpx.path_len[xe];
                        out.cpx.val_addr[xo] = out.cpx.val_addr[xe];
                        out.cpx.val_len[xo] = vl;
                    }
                    xcnt = w;
                    if (xcnt == 0 && xdm == INT64_MIN && xdl == LDT_NONE_U32) of &= ~PF_HAS_CPX;
                    else any_cell = true;
                }
                if (!(of & (PF_LIVE_TS | PF_ROW_DEL)) && !any_cell) of = 0;
                else if (pp.enforce_strict_liveness && !(of & PF_LIVE_TS) && !(of & PF_ROW_DEL)) of = 0;
            }
            if (of & PF_HAS_ROW) {
                if (sp.n_cpx) {
                    out.cpx_del_mfda[oslot] = xdm;
                    out.cpx_del_ldt[oslot] = xdl;
                    out.cpx_start[oslot] = xstart;
                    out.cpx_count[oslot] = (of & PF_HAS_CPX) ? xcnt : 0;
                }
                emit(BK_CLUSTERING, fck, of, lts, lttl, llet, rdm, rdl, INT64_MIN, LDT_NONE_U32,
                     true);
            }
        } else if (k == 1) {
            // single-version partition: UnfilteredRowIterators.merge of one
            // iterator returns it unchanged — markers pass through as-is and
            // only the Purger applies (PurgeFunction.applyToMarker).
            uint64_t o = mb[0] + mpos[0];
            uint8_t kd = in.rkind[o];
            int64_t e_m = in.rdel_mfda[o], s_m = in.start_mfda[o];
            uint32_t e_l = in.rdel_ldt[o], s_l = in.start_ldt[o];
            if (bk_is_boundary(kd)) {
                bool purge_close = purge_dt(e_m, e_l);
                bool purge_open = purge_dt(s_m, s_l);
                if (purge_close && purge_open) {
                } else if (purge_close) {
                    emit(kd == BK_EXCL_END_INCL_START ? BK_INCL_START : BK_EXCL_START, fck, 0,
                         NO_TIMESTAMP, 0, NO_DELETION_TIME, s_m, s_l, INT64_MIN, LDT_NONE_U32, false);
                } else if (purge_open) {
                    emit(kd == BK_EXCL_END_INCL_START ? BK_EXCL_END : BK_INCL_END, fck, 0,
                         NO_TIMESTAMP, 0, NO_DELETION_TIME, e_m, e_l, INT64_MIN, LDT_NONE_U32, false);
                } else {
                    emit(kd, fck, 0, NO_TIMESTAMP, 0, NO_DELETION_TIME, e_m, e_l, s_m, s_l, false);
                }
            } else {
                if (!purge_dt(e_m, e_l))
                    emit(kd, fck, 0, NO_TIMESTAMP, 0, NO_DELETION_TIME, e_m, e_l,
                         INT64_MIN, LDT_NONE_U32, false);
            }
        } else {
            // ---- marker event (RangeTombstoneMarker.Merger.merge) ----
            int64_t prev_m;
            uint32_t prev_l;
            merged_open(&prev_m, &prev_l);
            #pragma unroll
        for (uint32_t m = 0; m < (uint32_t)MA; m++) {
            if (m >= k) break;
                if (!(members & (1ULL << m))) continue;
                uint64_t o = mb[m] + mpos[m];
                uint8_t kd = in.rkind[o];
                if (bk_is_open(kd)) {
                    om_m[m] = bk_is_boundary(kd) ? in.start_mfda[o] : in.rdel_mfda[o];
                    om_l[m] = bk_is_boundary(kd) ? in.start_ldt[o] : in.rdel_ldt[o];
                    om_set |= 1ULL << m;
                } else {
                    om_set &= ~(1ULL << m);
                }
            }
            int64_t next_m;
            uint32_t next_l;
            merged_open(&next_m, &next_l);
            bool prev_live = prev_m == INT64_MIN && prev_l == LDT_NONE_U32;
            bool next_live = next_m == INT64_MIN && next_l == LDT_NONE_U32;
            if (!(prev_m == next_m && prev_l == next_l)) {
                // kind selection (RangeTombstoneMarker.java:124-147), reversed=false
                int ctc_tbl[8] = {-1, -1, -1, -1, 0, 1, 1, 1};
                bool before_clustering = ctc_tbl[fkind] < 0;
                uint8_t okind;
                int64_t e_m = INT64_MIN, s_m = INT64_MIN;
                uint32_t e_l = LDT_NONE_U32, s_l = LDT_NONE_U32;
                if (prev_live) {
                    okind = before_clustering ? BK_INCL_START : BK_EXCL_START;
                    e_m = next_m; e_l = next_l;
                } else if (next_live) {
                    okind = before_clustering ? BK_EXCL_END : BK_INCL_END;
                    e_m = prev_m; e_l = prev_l;
                } else {
                    okind = before_clustering ? BK_EXCL_END_INCL_START : BK_INCL_END_EXCL_START;
                    e_m = prev_m; e_l = prev_l;
                    s_m = next_m; s_l = next_l;
                }
                // ---- purge (PurgeFunction.applyToMarker, reversed=false) ----
                if (bk_is_boundary(okind)) {
                    bool purge_close = purge_dt(e_m, e_l);
                    bool purge_open = purge_dt(s_m, s_l);
                    if (purge_close && purge_open) {
                    } else if (purge_close) {
                        emit(okind == BK_EXCL_END_INCL_START ? BK_INCL_START : BK_EXCL_START, fck, 0,
                             NO_TIMESTAMP, 0, NO_DELETION_TIME, s_m, s_l, INT64_MIN, LDT_NONE_U32, false);
                    } else if (purge_open) {
                        emit(okind == BK_EXCL_END_INCL_START ? BK_EXCL_END : BK_INCL_END, fck, 0,
                             NO_TIMESTAMP, 0, NO_DELETION_TIME, e_m, e_l, INT64_MIN, LDT_NONE_U32, false);
                    } else {
                        emit(okind, fck, 0, NO_TIMESTAMP, 0, NO_DELETION_TIME, e_m, e_l, s_m, s_l, false);
                    }
                } else {
                    if (!purge_dt(e_m, e_l))
                        emit(okind, fck, 0, NO_TIMESTAMP, 0, NO_DELETION_TIME, e_m, e_l,
                             INT64_MIN, LDT_NONE_U32, false);
                }
            }
        }
        // advance members
        for (uint32_t m = 0; m < k; m++)
            if (members & (1ULL << m)) mpos[m]++;
    }
    op.row_count[g] = ocount;
    bool pdel_live = pdm == INT64_MIN && pdl == LDT_NONE_U32;
    op.pdel_mfda[g] = pdm;
    op.pdel_ldt[g] = pdl;
    op.keep[g] = (!pdel_live || ocount > 0 || static_kept) ? 1 : 0;
    if (op.keep[g]) {
        atomicMin(&st->first_group, (unsigned long long)g);
        atomicMax(&st->last_group, (unsigned long long)g);
    }
}

}  // namespace gpuc

namespace gpuc {

// ---------------------------------------------------------------------------
// serialization (SIZE pass then EMIT pass share one templated walk)
// ---------------------------------------------------------------------------
struct SerParams2 {
    HeaderStats hs;
    SchemaParams sch;
};

// one cell: serialized body size + flags byte (Cell.Serializer);
// COLS is UnfCols (regular cells) or StaticCols (same member names)
template <typename COLS>
__device__ inline uint32_t cell_body_size(const COLS& u, uint64_t oc, const SerParams2& sp,
                                          int32_t fixed, bool live, bool exp_live, uint64_t o,
                                          uint8_t* out_cf) {
    uint8_t f = u.cell_flags[oc];
    int64_t cts = u.cell_ts[oc];
    uint32_t cldt = u.cell_ldt[oc];
    int32_t cttl = u.cell_ttl[oc];
    uint32_t vlen = u.val_len[oc];
    bool has_value = vlen > 0 && (f & CELLF_HAS_VALUE);
    bool deleted = cldt != LDT_NONE_U32 && cttl == NO_TTL;
    bool expiring = cttl != NO_TTL;
    bool use_row_ts = live && cts == u.live_ts[o];
    bool use_row_ttl = expiring && exp_live && cttl == u.live_ttl[o] && ldt_long(cldt) == u.live_let[o];
    uint8_t cflags = 0;
    if (!has_value) cflags |= 4;
    if (deleted) cflags |= 1;
    else if (expiring) cflags |= 2;
    if (use_row_ts) cflags |= 8;
    if (use_row_ttl) cflags |= 16;
    uint32_t body = 1;
    if (!use_row_ts) body += uvint_size((uint64_t)(cts - sp.hs.min_ts));
    if ((deleted || expiring) && !use_row_ttl) body += uvint_size(sext32(ldt_long(cldt) - sp.hs.min_ldt));
    if (expiring && !use_row_ttl) body += uvint_size(sext32(cttl - sp.hs.min_ttl));
    if (has_value) body += (fixed >= 0 ? 0 : uvint_size(vlen)) + vlen;
    *out_cf = cflags;
    return body;
}

// complex-cell body size (Cell.Serializer + CellPath vint+bytes; value is
// the map VALUE type == BytesType, so always vint-length)
__device__ inline uint32_t cpx_cell_body_size(const UnfCols& u, uint64_t xe, const SerParams2& sp,
                                              bool live, bool exp_live, uint64_t o,
                                              uint8_t* out_cf) {
    uint8_t f = u.cpx.flags[xe];
    int64_t cts = u.cpx.ts[xe];
    uint32_t cldt = u.cpx.ldt[xe];
    int32_t cttl = u.cpx.ttl[xe];
    uint32_t vlen = u.cpx.val_len[xe];
    bool has_value = vlen > 0 && (f & CELLF_HAS_VALUE);
    bool deleted = cldt != LDT_NONE_U32 && cttl == NO_TTL;
    bool expiring = cttl != NO_TTL;
    bool use_row_ts = live && cts == u.live_ts[o];
    bool use_row_ttl = expiring && exp_live && cttl == u.live_ttl[o] && ldt_long(cldt) == u.live_let[o];
    uint8_t cflags = 0;
    if (!has_value) cflags |= 4;
    if (deleted) cflags |= 1;
    else if (expiring) cflags |= 2;
    if (use_row_ts) cflags |= 8;
    if (use_row_ttl) cflags |= 16;
    uint32_t body = 1;
    if (!use_row_ts) body += uvint_size((uint64_t)(cts - sp.hs.min_ts));
    if ((deleted || expiring) && !use_row_ttl) body += uvint_size(sext32(ldt_long(cldt) - sp.hs.min_ldt));
    if (expiring && !use_row_ttl) body += uvint_size(sext32(cttl - sp.hs.min_ttl));
    uint32_t plen = u.cpx.path_len[xe];
    body += uvint_size(plen) + plen;
    if (has_value) body += uvint_size(vlen) + vlen;
    if (out_cf) *out_cf = cflags;
    return body;
}

// row/marker body size + flag bytes (mirrors the oracle serializers)
__device__ inline uint32_t unf_body_size(const UnfCols& u, uint64_t o, const SerParams2& sp,
                                         uint8_t* out_flags, uint8_t* out_cflags) {
    uint8_t kind = u.rkind[o];
    if (kind != BK_CLUSTERING) {
        uint32_t body = uvint_size((uint64_t)(u.rdel_mfda[o] - sp.hs.min_ts)) +
                        uvint_size(sext32(ldt_long(u.rdel_ldt[o]) - sp.hs.min_ldt));
        if (bk_is_boundary(kind))
            body += uvint_size((uint64_t)(u.start_mfda[o] - sp.hs.min_ts)) +
                    uvint_size(sext32(ldt_long(u.start_ldt[o]) - sp.hs.min_ldt));
        if (out_flags) *out_flags = 0x02;
        if (out_cflags) *out_cflags = 0;
        return body;
    }
    uint8_t f = u.flags[o];
    uint32_t body = 0;
    uint8_t rflags = 0;
    bool live = f & PF_LIVE_TS;
    bool exp_live = live && u.live_ttl[o] != NO_TTL;
    if (live) { rflags |= 0x04; body += uvint_size((uint64_t)(u.live_ts[o] - sp.hs.min_ts)); }
    if (exp_live) {
        rflags |= 0x08;
        body += uvint_size(sext32(u.live_ttl[o] - sp.hs.min_ttl));
        body += uvint_size(sext32(u.live_let[o] - sp.hs.min_ldt));
    }
    if (f & PF_ROW_DEL) {
        rflags |= 0x10;
        body += uvint_size((uint64_t)(u.rdel_mfda[o] - sp.hs.min_ts));
        body += uvint_size(sext32(ldt_long(u.rdel_ldt[o]) - sp.hs.min_ldt));
    }
    // columns subset + per-cell sizes (cell_flags_byte shared with EMIT);
    // the superset covers simple columns + the complex column (bit n_cols)
    uint64_t present_mask = 0;
    uint32_t present = 0;
    const uint32_t NSUP = sp.sch.n_cols + sp.sch.n_cpx;
    bool has_cpx = sp.sch.n_cpx && (f & PF_HAS_CPX);
    for (uint32_t c = 0; c < sp.sch.n_cols; c++)
        if (u.cell_flags[o * sp.sch.n_cols + c] & CELLF_PRESENT) { present_mask |= 1ULL << c; present++; }
    if (has_cpx) { present_mask |= 1ULL << sp.sch.n_cols; present++; }
    if (present == NSUP) rflags |= 0x20;
    else body += uvint_size(((1ULL << NSUP) - 1) & ~present_mask);
    for (uint32_t c = 0; c < sp.sch.n_cols; c++) {
        uint64_t oc = o * sp.sch.n_cols + c;
        if (!(u.cell_flags[oc] & CELLF_PRESENT)) continue;
        uint8_t cflags;
        body += cell_body_size(u, oc, sp, sp.sch.col_fixed[c], live, exp_live, o, &cflags);
    }
    if (has_cpx) {
        bool del_live = u.cpx_del_mfda[o] == INT64_MIN && u.cpx_del_ldt[o] == LDT_NONE_U32;
        if (!del_live) rflags |= 0x40;  // HAS_COMPLEX_DELETION
        if (rflags & 0x40)
            body += uvint_size((uint64_t)(u.cpx_del_mfda[o] - sp.hs.min_ts)) +
                    uvint_size(sext32(ldt_long(u.cpx_del_ldt[o]) - sp.hs.min_ldt));
        uint32_t nc2 = u.cpx_count[o];
        body += uvint_size(nc2);
        uint64_t si = u.cpx_start[o];
        for (uint32_t e2 = 0; e2 < nc2; e2++)
            body += cpx_cell_body_size(u, si + e2, sp, live, exp_live, o, nullptr);
    }
    if (out_flags) *out_flags = rflags;
    if (out_cflags) *out_cflags = 0;
    return body;
}

// the static row: size (always) and bytes (when out_data != nullptr) at
// data_off+start. Written whenever the schema has statics — possibly as an
// EMPTY static row (flags 0x80, ext 0x01, all-missing subset), exactly as
// SortedTablePartitionWriter.addStaticRow does.
__device__ inline uint64_t static_row_bytes(const OutParts& op, uint64_t g,
                                            const SerParams2& sp, uint8_t* out_data,
                                            uint64_t data_off, uint64_t start, int lane) {
    const uint32_t NS = sp.sch.n_static;
    const StaticCols& st = op.st;
    uint8_t sof = st.flags[g];
    bool live = sof & PF_LIVE_TS;
    bool exp_live = live && st.live_ttl[g] != NO_TTL;
    bool has_del = sof & PF_ROW_DEL;
    uint64_t pmask = 0;
    uint32_t present = 0;
    for (uint32_t c = 0; c < NS; c++)
        if (st.cell_flags[g * NS + c] & CELLF_PRESENT) { pmask |= 1ULL << c; present++; }
    uint8_t rf = 0x80;
    if (live) rf |= 0x04;
    if (exp_live) rf |= 0x08;
    if (has_del) rf |= 0x10;
    if (present == NS) rf |= 0x20;
    uint32_t body = 0;
    if (live) body += uvint_size((uint64_t)(st.live_ts[g] - sp.hs.min_ts));
    if (exp_live) body += uvint_size(sext32(st.live_ttl[g] - sp.hs.min_ttl)) +
                          uvint_size(sext32(st.live_let[g] - sp.hs.min_ldt));
    if (has_del) body += uvint_size((uint64_t)(st.rdel_mfda[g] - sp.hs.min_ts)) +
                         uvint_size(sext32(ldt_long(st.rdel_ldt[g]) - sp.hs.min_ldt));
    uint64_t miss = ((NS >= 64 ? ~0ULL : ((1ULL << NS) - 1)) & ~pmask);
    if (!(rf & 0x20)) body += uvint_size(miss);
    uint8_t cfb_arr[63];
    for (uint32_t c = 0; c < NS; c++) {
        if (!(pmask & (1ULL << c))) continue;
        body += cell_body_size(st, g * NS + c, sp, sp.sch.static_fixed[c], live, exp_live, g,
                               &cfb_arr[c]);
    }
    uint64_t total = 2 + uvint_size(body + uvint_size(0)) + uvint_size(0) + body;
    if (!out_data) return total;
    // ---- emit ----
    uint64_t pos = start;
    auto put8 = [&](uint8_t v) { if (lane == 0) out_data[data_off + pos] = v; pos++; };
    auto put_uv = [&](uint64_t v) {
        if (lane == 0) {
            uint8_t tmp[9];
            int n = uvint_put(tmp, v);
            for (int i = 0; i < n; i++) out_data[data_off + pos + i] = tmp[i];
        }
        pos += uvint_size(v);
    };
    put8(rf);
    put8(0x01);  // extended flags: IS_STATIC
    put_uv(body + uvint_size(0));
    put_uv(0);   // previousUnfilteredSize
    if (live) put_uv((uint64_t)(st.live_ts[g] - sp.hs.min_ts));
    if (exp_live) {
        put_uv(sext32(st.live_ttl[g] - sp.hs.min_ttl));
        put_uv(sext32(st.live_let[g] - sp.hs.min_ldt));
    }
    if (has_del) {
        put_uv((uint64_t)(st.rdel_mfda[g] - sp.hs.min_ts));
        put_uv(sext32(ldt_long(st.rdel_ldt[g]) - sp.hs.min_ldt));
    }
    if (!(rf & 0x20)) put_uv(miss);
    for (uint32_t c = 0; c < NS; c++) {
        if (!(pmask & (1ULL << c))) continue;
        uint64_t oc = g * NS + c;
        uint8_t cfb = cfb_arr[c];
        put8(cfb);
        if (!(cfb & 8)) put_uv((uint64_t)(st.cell_ts[oc] - sp.hs.min_ts));
        bool deleted = cfb & 1, expiring = cfb & 2;
        if ((deleted || expiring) && !(cfb & 16))
            put_uv(sext32(ldt_long(st.cell_ldt[oc]) - sp.hs.min_ldt));
        if (expiring && !(cfb & 16)) put_uv(sext32(st.cell_ttl[oc] - sp.hs.min_ttl));
        if (!(cfb & 4)) {
            uint32_t vlen = st.val_len[oc];
            if (sp.sch.static_fixed[c] < 0) put_uv(vlen);
            wave_copy(out_data + data_off + pos, (const uint8_t*)st.val_addr[oc],
                      (int)vlen, lane);
            pos += vlen;
        }
    }
    return total;
}

// serialized size of a ClusteringPrefix (full serializer: kind byte [+ u16
// size for bounds] + values-without-size) — IndexInfo first/last names
__device__ inline uint32_t prefix_full_size(const UnfCols& u, uint64_t o,
                                            const SchemaParams& sch) {
    uint8_t kind = u.rkind[o];
    uint32_t nv = u.ck_count[o];
    uint32_t s = 1;                       // kind byte
    if (kind != BK_CLUSTERING) s += 2;    // u16 value count
    if (nv) s += 1;                       // 32-batch header vint (all non-null)
    for (uint32_t c = 0; c < nv; c++) {
        if (sch.ck_w[c] > 0) s += sch.ck_w[c];
        else {
            uint32_t vlen = u.ck_len[o * sch.n_ck + c];
            s += uvint_size(vlen) + vlen;
        }
    }
    return s;
}
__device__ inline uint32_t dt_ser_size(int64_t m, uint32_t l) {
    return (m == INT64_MIN && l == LDT_NONE_U32) ? 1 : 12;
}

// one partition: Data bytes (+ IndexInfo bookkeeping). EMIT=false computes
// sizes only (thread, lane==0); EMIT=true writes bytes (wave; lane 0 meta,
// all lanes value copies). Returns psize; outputs isize/nblocks/infos_size.
template <bool EMIT>
__device__ uint64_t part_walk(const OutParts& op, const UnfCols& out, uint64_t g,
                              const SerParams2& sp, uint64_t data_off, uint64_t idx_off,
                              uint64_t infos_size_hint, uint32_t nblocks_hint,
                              uint8_t* out_data, uint8_t* out_index, int lane,
                              uint64_t* isize_out, uint32_t* nblocks_out,
                              uint64_t* infos_size_out) {
    if (!op.keep[g]) {
        if (isize_out) *isize_out = 0;
        if (nblocks_out) *nblocks_out = 0;
        if (infos_size_out) *infos_size_out = 0;
        return 0;
    }
    uint32_t klen = op.klen[g];
    const uint8_t* keyb = (const uint8_t*)op.key_addr[g];
    bool pdel_live = op.pdel_mfda[g] == INT64_MIN && op.pdel_ldt[g] == LDT_NONE_U32;
    uint64_t static_sz = sp.sch.n_static
                             ? static_row_bytes(op, g, sp, nullptr, 0, 0, lane)
                             : 0;
    uint64_t header_len = 2 + klen + (pdel_live ? 1 : 12) + static_sz;
    uint64_t rb = op.row_base[g];
    uint32_t nrows = op.row_count[g];

    uint64_t pos = 0;
    auto emit8 = [&](uint8_t v) {
        if (EMIT && lane == 0) out_data[data_off + pos] = v;
        pos++;
    };
    auto emit_uv = [&](uint64_t v) {
        if (EMIT && lane == 0) {
            uint8_t tmp[9];
            int n = uvint_put(tmp, v);
            for (int i = 0; i < n; i++) out_data[data_off + pos + i] = tmp[i];
            pos += n;
        } else pos += uvint_size(v);
    };
    auto emit_dt = [&](int64_t m, uint32_t l) {
        if (m == INT64_MIN && l == LDT_NONE_U32) { emit8(0x80); return; }
        if (EMIT && lane == 0) {
            for (int i = 0; i < 8; i++) out_data[data_off + pos + i] = (uint8_t)((uint64_t)m >> (8 * (7 - i)));
            for (int i = 0; i < 4; i++) out_data[data_off + pos + 8 + i] = (uint8_t)(l >> (8 * (3 - i)));
        }
        pos += 12;
    };
    auto emit_ck = [&](uint64_t o) {
        uint32_t nv = out.ck_count[o];
        if (nv == 0) return;
        if (EMIT && lane == 0) out_data[data_off + pos] = 0;  // header: all non-null
        pos += 1;
        for (uint32_t c = 0; c < nv; c++) {
            uint64_t oc = o * sp.sch.n_ck + c;
            if (sp.sch.ck_w[c] > 0) {
                if (EMIT && lane == 0) ck_bytes(out.ck[oc], sp.sch.ck_w[c], &out_data[data_off + pos]);
                pos += sp.sch.ck_w[c];
            } else {
                uint32_t vlen = out.ck_len[oc];
                if (EMIT && lane == 0) {
                    uint8_t tmp[9];
                    int n = uvint_put(tmp, vlen);
                    for (int i = 0; i < n; i++) out_data[data_off + pos + i] = tmp[i];
                    const uint8_t* src = (const uint8_t*)out.ck_addr[oc];
                    for (uint32_t i = 0; i < vlen; i++) out_data[data_off + pos + n + i] = src[i];
                }
                pos += uvint_size(vlen) + vlen;
            }
        }
    };

    // ---- index-entry layout (EMIT uses the SIZE pass results) ----
    uint64_t fields_size = uvint_size(header_len) + dt_ser_size(op.pdel_mfda[g], op.pdel_ldt[g]) +
                           uvint_size(nblocks_hint);
    uint64_t S_hint = fields_size + infos_size_hint + (uint64_t)nblocks_hint * 4;
    uint64_t infos_begin = 0, offsets_begin = 0;
    if (EMIT && nblocks_hint > 1) {
        infos_begin = idx_off + 2 + klen + uvint_size(data_off) + uvint_size(S_hint) + fields_size;
        offsets_begin = infos_begin + infos_size_hint;
    }

    // ---- partition header ----
    emit8((uint8_t)(klen >> 8));
    emit8((uint8_t)klen);
    for (uint32_t b = 0; b < klen; b++) emit8(keyb[b]);
    emit_dt(op.pdel_mfda[g], op.pdel_ldt[g]);
    if (sp.sch.n_static) {
        if (EMIT) static_row_bytes(op, g, sp, out_data, data_off, pos, lane);
        pos += static_sz;
    }

    // ---- walk unfiltereds ----
    uint64_t prev_start = 0;
    bool block_open = false;
    uint64_t block_start = 0;
    uint64_t block_first_o = 0, last_o = 0;  // row indices (kind/ck read back)
    int64_t open_m = INT64_MIN;
    uint32_t open_l = LDT_NONE_U32;
    uint32_t nblocks = 0;
    uint64_t infos_size = 0;
    uint64_t info_cursor = infos_begin;

    auto emit_prefix_idx = [&](uint64_t o, uint64_t q) -> uint64_t {
        uint8_t kk = out.rkind[o];
        uint32_t nv = out.ck_count[o];
        if (EMIT && lane == 0) out_index[q] = kk;
        q++;
        if (kk != BK_CLUSTERING) {
            if (EMIT && lane == 0) { out_index[q] = 0; out_index[q + 1] = (uint8_t)nv; }
            q += 2;
        }
        if (nv) {
            if (EMIT && lane == 0) out_index[q] = 0;
            q += 1;
            for (uint32_t c = 0; c < nv; c++) {
                uint64_t oc = o * sp.sch.n_ck + c;
                if (sp.sch.ck_w[c] > 0) {
                    if (EMIT && lane == 0) ck_bytes(out.ck[oc], sp.sch.ck_w[c], &out_index[q]);
                    q += sp.sch.ck_w[c];
                } else {
                    uint32_t vlen = out.ck_len[oc];
                    if (EMIT && lane == 0) {
                        uint8_t tmp[9];
                        int n = uvint_put(tmp, vlen);
                        for (int i = 0; i < n; i++) out_index[q + i] = tmp[i];
                        const uint8_t* src = (const uint8_t*)out.ck_addr[oc];
                        for (uint32_t i = 0; i < vlen; i++) out_index[q + n + i] = src[i];
                    }
                    q += uvint_size(vlen) + vlen;
                }
            }
        }
        return q;
    };
    auto flush_block = [&](uint64_t end_pos) {
        uint64_t width = end_pos - block_start;
        bool has_open = !(open_m == INT64_MIN && open_l == LDT_NONE_U32);
        uint64_t isz = prefix_full_size(out, block_first_o, sp.sch) +
                       prefix_full_size(out, last_o, sp.sch) +
                       uvint_size(block_start) + uvint_size(zigzag((int64_t)width - 65536)) + 1 +
                       (has_open ? dt_ser_size(open_m, open_l) : 0);
        if (EMIT && nblocks_hint > 1) {
            if (lane == 0) {
                uint64_t rel = info_cursor - infos_begin;
                uint64_t slot = offsets_begin + (uint64_t)nblocks * 4;
                out_index[slot] = (uint8_t)(rel >> 24);
                out_index[slot + 1] = (uint8_t)(rel >> 16);
                out_index[slot + 2] = (uint8_t)(rel >> 8);
                out_index[slot + 3] = (uint8_t)rel;
            }
            uint64_t q = emit_prefix_idx(block_first_o, info_cursor);
            q = emit_prefix_idx(last_o, q);
            if (lane == 0) {
                uint8_t tmp[10];
                int n = uvint_put(tmp, block_start);
                for (int i = 0; i < n; i++) out_index[q + i] = tmp[i];
            }
            q += uvint_size(block_start);
            if (lane == 0) {
                uint8_t tmp[10];
                int n = uvint_put(tmp, zigzag((int64_t)width - 65536));
                for (int i = 0; i < n; i++) out_index[q + i] = tmp[i];
            }
            q += uvint_size(zigzag((int64_t)width - 65536));
            if (lane == 0) out_index[q] = has_open ? 1 : 0;
            q += 1;
            if (has_open) {
                if (lane == 0) {
                    for (int i = 0; i < 8; i++) out_index[q + i] = (uint8_t)((uint64_t)open_m >> (8 * (7 - i)));
                    for (int i = 0; i < 4; i++) out_index[q + 8 + i] = (uint8_t)(open_l >> (8 * (3 - i)));
                }
                q += 12;
            }
            info_cursor = q;
        }
        infos_size += isz;
        nblocks++;
        block_open = false;
    };

    for (uint32_t j = 0; j < nrows; j++) {
        uint64_t o = rb + j;
        uint64_t upos = pos;
        if (!block_open) {
            block_open = true;
            block_start = upos;
            block_first_o = o;
        }
        uint8_t kind = out.rkind[o];
        uint8_t rflags = 0, cflags = 0;
        uint32_t body = unf_body_size(out, o, sp, &rflags, &cflags);
        uint64_t prev_sz = j == 0 ? header_len : upos - prev_start;
        if (kind != BK_CLUSTERING) {
            emit8(0x02);
            emit8(kind);
            emit8(0);
            emit8(out.ck_count[o]);
            emit_ck(o);
            emit_uv(body + uvint_size(prev_sz));
            emit_uv(prev_sz);
            emit_uv((uint64_t)(out.rdel_mfda[o] - sp.hs.min_ts));
            emit_uv(sext32(ldt_long(out.rdel_ldt[o]) - sp.hs.min_ldt));
            if (bk_is_boundary(kind)) {
                emit_uv((uint64_t)(out.start_mfda[o] - sp.hs.min_ts));
                emit_uv(sext32(ldt_long(out.start_ldt[o]) - sp.hs.min_ldt));
            }
            open_m = bk_is_open(kind) ? (bk_is_boundary(kind) ? out.start_mfda[o] : out.rdel_mfda[o]) : INT64_MIN;
            open_l = bk_is_open(kind) ? (bk_is_boundary(kind) ? out.start_ldt[o] : out.rdel_ldt[o]) : LDT_NONE_U32;
        } else {
            emit8(rflags);
            emit_ck(o);
            emit_uv(body + uvint_size(prev_sz));
            emit_uv(prev_sz);
            if (rflags & 0x04) emit_uv((uint64_t)(out.live_ts[o] - sp.hs.min_ts));
            if (rflags & 0x08) {
                emit_uv(sext32(out.live_ttl[o] - sp.hs.min_ttl));
                emit_uv(sext32(out.live_let[o] - sp.hs.min_ldt));
            }
            if (rflags & 0x10) {
                emit_uv((uint64_t)(out.rdel_mfda[o] - sp.hs.min_ts));
                emit_uv(sext32(ldt_long(out.rdel_ldt[o]) - sp.hs.min_ldt));
            }
            if (!(rflags & 0x20)) {
                uint64_t present_mask = 0;
                const uint32_t NSUP = sp.sch.n_cols + sp.sch.n_cpx;
                for (uint32_t c = 0; c < sp.sch.n_cols; c++)
                    if (out.cell_flags[o * sp.sch.n_cols + c] & CELLF_PRESENT) present_mask |= 1ULL << c;
                if (sp.sch.n_cpx && (out.flags[o] & PF_HAS_CPX)) present_mask |= 1ULL << sp.sch.n_cols;
                emit_uv(((1ULL << NSUP) - 1) & ~present_mask);
            }
            bool live = rflags & 0x04, exp_live = rflags & 0x08;
            for (uint32_t c = 0; c < sp.sch.n_cols; c++) {
                uint64_t oc = o * sp.sch.n_cols + c;
                if (!(out.cell_flags[oc] & CELLF_PRESENT)) continue;
                uint8_t cfb;
                cell_body_size(out, oc, sp, sp.sch.col_fixed[c], live, exp_live, o, &cfb);
                emit8(cfb);
                if (!(cfb & 8)) emit_uv((uint64_t)(out.cell_ts[oc] - sp.hs.min_ts));
                bool deleted = cfb & 1, expiring = cfb & 2;
                if ((deleted || expiring) && !(cfb & 16))
                    emit_uv(sext32(ldt_long(out.cell_ldt[oc]) - sp.hs.min_ldt));
                if (expiring && !(cfb & 16)) emit_uv(sext32(out.cell_ttl[oc] - sp.hs.min_ttl));
                if (!(cfb & 4)) {
                    uint32_t vlen = out.val_len[oc];
                    if (sp.sch.col_fixed[c] < 0) emit_uv(vlen);
                    if (EMIT)
                        wave_copy(out_data + data_off + pos,
                                  (const uint8_t*)out.val_addr[oc], (int)vlen, lane);
                    pos += vlen;
                }
            }
            // ---- complex column (UnfilteredSerializer.writeComplexColumn) ----
            if (sp.sch.n_cpx && (out.flags[o] & PF_HAS_CPX)) {
                if (rflags & 0x40) {
                    emit_uv((uint64_t)(out.cpx_del_mfda[o] - sp.hs.min_ts));
                    emit_uv(sext32(ldt_long(out.cpx_del_ldt[o]) - sp.hs.min_ldt));
                }
                uint32_t nc2 = out.cpx_count[o];
                emit_uv(nc2);
                uint64_t si = out.cpx_start[o];
                for (uint32_t e2 = 0; e2 < nc2; e2++) {
                    uint64_t xe = si + e2;
                    uint8_t cfb;
                    cpx_cell_body_size(out, xe, sp, live, exp_live, o, &cfb);
                    emit8(cfb);
                    if (!(cfb & 8)) emit_uv((uint64_t)(out.cpx.ts[xe] - sp.hs.min_ts));
                    bool deleted = cfb & 1, expiring = cfb & 2;
                    if ((deleted || expiring) && !(cfb & 16))
                        emit_uv(sext32(ldt_long(out.cpx.ldt[xe]) - sp.hs.min_ldt));
                    if (expiring && !(cfb & 16)) emit_uv(sext32(out.cpx.ttl[xe] - sp.hs.min_ttl));
                    uint32_t plen = out.cpx.path_len[xe];
                    emit_uv(plen);
                    if (EMIT)
                        wave_copy(out_data + data_off + pos,
                                  (const uint8_t*)out.cpx.path_addr[xe], (int)plen, lane);
                    pos += plen;
                    if (!(cfb & 4)) {
                        uint32_t vlen = out.cpx.val_len[xe];
                        emit_uv(vlen);
                        if (EMIT)
                            wave_copy(out_data + data_off + pos,
                                      (const uint8_t*)out.cpx.val_addr[xe], (int)vlen, lane);
                        pos += vlen;
                    }
                }
            }
        }
        prev_start = upos;
        last_o = o;
        if (pos - block_start >= sp.sch.column_index_size) flush_block(pos);
    }
    emit8(0x01);  // END_OF_PARTITION
    // the final block's width includes the end byte (addIndexBlock runs after
    // writeEndOfPartition in BigFormatPartitionWriter.finish)
    if (nrows > 0 && block_open) flush_block(pos);

    // ---- index entry ----
    uint64_t entry = 2 + klen + uvint_size(data_off);
    if (nblocks > 1) {
        uint64_t S = fields_size + infos_size + (uint64_t)nblocks * 4;
        entry += uvint_size(S) + S;
        if (EMIT) {
            uint64_t q = idx_off;
            if (lane == 0) {
                out_index[q] = (uint8_t)(klen >> 8);
                out_index[q + 1] = (uint8_t)klen;
                for (uint32_t b = 0; b < klen; b++) out_index[q + 2 + b] = keyb[b];
            }
            q += 2 + klen;
            uint8_t tmp[10];
            if (lane == 0) {
                int n = uvint_put(tmp, data_off);
                for (int i = 0; i < n; i++) out_index[q + i] = tmp[i];
            }
            q += uvint_size(data_off);
            if (lane == 0) {
                int n = uvint_put(tmp, S);
                for (int i = 0; i < n; i++) out_index[q + i] = tmp[i];
            }
            q += uvint_size(S);
            if (lane == 0) {
                int n = uvint_put(tmp, header_len);
                for (int i = 0; i < n; i++) out_index[q + i] = tmp[i];
            }
            q += uvint_size(header_len);
            if (pdel_live) {
                if (lane == 0) out_index[q] = 0x80;
                q += 1;
            } else {
                if (lane == 0) {
                    for (int i = 0; i < 8; i++) out_index[q + i] = (uint8_t)((uint64_t)op.pdel_mfda[g] >> (8 * (7 - i)));
                    for (int i = 0; i < 4; i++) out_index[q + 8 + i] = (uint8_t)(op.pdel_ldt[g] >> (8 * (3 - i)));
                }
                q += 12;
            }
            if (lane == 0) {
                int n = uvint_put(tmp, nblocks);
                for (int i = 0; i < n; i++) out_index[q + i] = tmp[i];
            }
            q += uvint_size(nblocks);
            // infos + offsets were written during the walk at infos_begin/offsets_begin
        }
    } else {
        entry += 1;  // vint promoted-size 0 (RowIndexEntry.java:468-473)
        if (EMIT && lane == 0) {
            uint64_t q = idx_off;
            out_index[q] = (uint8_t)(klen >> 8);
            out_index[q + 1] = (uint8_t)klen;
            for (uint32_t b = 0; b < klen; b++) out_index[q + 2 + b] = keyb[b];
            q += 2 + klen;
            uint8_t tmp[10];
            int n = uvint_put(tmp, data_off);
            for (int i = 0; i < n; i++) out_index[q + i] = tmp[i];
            q += n;
            out_index[q] = 0;
        }
    }
    if (isize_out) *isize_out = entry;
    if (nblocks_out) *nblocks_out = nblocks;
    if (infos_size_out) *infos_size_out = infos_size;
    return pos;
}

// SIZE pass: psize/isize (+nblocks, infos_size) + output-stats histograms
__global__ void k_sizes_rows(OutParts op, UnfCols out, uint64_t n, SerParams2 sp,
                             uint64_t* psize, uint64_t* isize, uint32_t* nblocks,
                             uint64_t* infos_size, OutStats* st,
                             const int64_t* ps_hist_off, int32_t ps_hist_n,
                             const int64_t* ch_hist_off, int32_t ch_hist_n) {
    __shared__ unsigned int sh_ps[156], sh_ch[119];
    for (int i = threadIdx.x; i < 156; i += blockDim.x) sh_ps[i] = 0;
    for (int i = threadIdx.x; i < 119; i += blockDim.x) sh_ch[i] = 0;
    __syncthreads();
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g < n) {
        uint64_t is, inf;
        uint32_t nb;
        uint64_t s = part_walk<false>(op, out, g, sp, 0, 0, 0, 0, nullptr, nullptr, 0, &is, &nb, &inf);
        psize[g] = s;
        isize[g] = 0;  // computed by k_index_sizes_rows once data offsets exist
        nblocks[g] = nb;
        infos_size[g] = inf;
        if (s) {
            int lo = 0, hi = ps_hist_n;
            while (lo < hi) { int mid = (lo + hi) >> 1; if ((uint64_t)ps_hist_off[mid] < s) lo = mid + 1; else hi = mid; }
            atomicAdd(&sh_ps[lo], 1u);
            uint64_t cells = 0;
            for (uint32_t j = 0; j < op.row_count[g]; j++) {
                uint64_t ro = op.row_base[g] + j;
                for (uint32_t c = 0; c < sp.sch.n_cols; c++)
                    if (out.cell_flags[ro * sp.sch.n_cols + c] & CELLF_PRESENT) cells++;
                if (sp.sch.n_cpx && (out.flags[ro] & PF_HAS_CPX)) cells += out.cpx_count[ro];
            }
            lo = 0; hi = ch_hist_n;
            while (lo < hi) { int mid = (lo + hi) >> 1; if ((uint64_t)ch_hist_off[mid] < cells) lo = mid + 1; else hi = mid; }
            atomicAdd(&sh_ch[lo], 1u);
        }
    }
    __syncthreads();
    for (int i = threadIdx.x; i < 156; i += blockDim.x)
        if (sh_ps[i]) atomicAdd(&st->part_size_hist[i], (unsigned long long)sh_ps[i]);
    for (int i = threadIdx.x; i < 119; i += blockDim.x)
        if (sh_ch[i]) atomicAdd(&st->cells_hist[i], (unsigned long long)sh_ch[i]);
}

// index entry sizes — needs the scanned Data offsets (the position vint's
// width depends on the partition's absolute position)
__global__ void k_index_sizes_rows(OutParts op, uint64_t n, SerParams2 sp,
                                   const uint64_t* data_off,
                                   const uint32_t* nblocks, const uint64_t* infos_size,
                                   uint64_t* isize) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n) return;
    if (!op.keep[g]) { isize[g] = 0; return; }
    uint32_t klen = op.klen[g];
    bool pdel_live = op.pdel_mfda[g] == INT64_MIN && op.pdel_ldt[g] == LDT_NONE_U32;
    // header_len INCLUDES the static row (headerLength is taken after
    // addStaticRow in SortedTablePartitionWriter) — keep in sync with
    // part_walk's computation
    uint64_t header_len = 2 + klen + (pdel_live ? 1 : 12) +
                          (sp.sch.n_static ? static_row_bytes(op, g, sp, nullptr, 0, 0, 0) : 0);
    uint64_t e = 2 + klen + uvint_size(data_off[g]);
    uint32_t nb = nblocks[g];
    if (nb > 1) {
        uint64_t S = uvint_size(header_len) + dt_ser_size(op.pdel_mfda[g], op.pdel_ldt[g]) +
                     uvint_size(nb) + infos_size[g] + (uint64_t)nb * 4;
        e += uvint_size(S) + S;
    } else {
        e += 1;
    }
    isize[g] = e;
}

// EMIT pass: wave per partition (Data + Index + bloom)
__global__ void k_serialize_rows(OutParts op, UnfCols out, uint64_t n, SerParams2 sp,
                                 const uint64_t* data_off, const uint64_t* idx_off,
                                 const uint32_t* nblocks, const uint64_t* infos_size,
                                 uint8_t* out_data, uint8_t* out_index,
                                 uint32_t* bloom_bits, uint64_t bloom_bitlen, int32_t bloom_k,
                                 uint64_t g0 = 0) {
    uint64_t g = g0 + blockIdx.x * (blockDim.x / WAVE) + (threadIdx.x / WAVE);
    if (g >= n || !op.keep[g]) return;
    int lane = threadIdx.x & (WAVE - 1);
    part_walk<true>(op, out, g, sp, data_off[g], idx_off[g], infos_size[g], nblocks[g],
                    out_data, out_index, lane, nullptr, nullptr, nullptr);
    if (lane == 0) {
        uint32_t klen = op.klen[g];
        uint64_t h[2];
        murmur3_128((const uint8_t*)op.key_addr[g], klen, 0, h);
        int64_t base = (int64_t)h[1], inc = (int64_t)h[0];
        for (int i = 0; i < bloom_k; i++) {
            int64_t m = base % (int64_t)bloom_bitlen;
            uint64_t idx = (uint64_t)((m ^ (m >> 63)) - (m >> 63));
            atomicOr(&bloom_bits[idx >> 5], 1u << (idx & 31));
            base += inc;
        }
    }
}

// output stats over the final OutParts/UnfCols (MetadataCollector semantics)
__global__ void k_collect_rows(OutParts op, UnfCols out, uint64_t n, uint32_t n_cols, OutStats* st,
                               uint32_t* tomb_ldts, uint32_t tomb_cap) {
    __shared__ unsigned long long sh_parts, sh_rows, sh_cells, sh_mints, sh_maxts,
        sh_minldt, sh_maxldt, sh_haspdel;
    __shared__ unsigned int sh_minttl, sh_maxttl;
    if (threadIdx.x == 0) {
        sh_parts = sh_rows = sh_cells = 0;
        sh_mints = sh_minldt = 0xFFFFFFFFFFFFFFFFULL;
        sh_maxts = sh_maxldt = 0;
        sh_haspdel = 0;
        sh_minttl = 0xFFFFFFFFu;
        sh_maxttl = 0;
    }
    __syncthreads();
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g < n && op.keep[g]) {
        auto lts = [&](int64_t ts) {
            if (ts == NO_TIMESTAMP) return;
            unsigned long long v = (unsigned long long)ts ^ 0x8000000000000000ULL;
            atomicMin(&sh_mints, v);
            atomicMax(&sh_maxts, v);
        };
        auto lldt = [&](int64_t l) {
            unsigned long long v = (unsigned long long)l ^ 0x8000000000000000ULL;
            atomicMin(&sh_minldt, v);
            atomicMax(&sh_maxldt, v);
        };
        atomicAdd(&sh_parts, 1ull);
        bool pdel_live = op.pdel_mfda[g] == INT64_MIN && op.pdel_ldt[g] == LDT_NONE_U32;
        if (!pdel_live) {
            atomicExch(&sh_haspdel, 1ull);
            lts(op.pdel_mfda[g]);
            lldt(ldt_long(op.pdel_ldt[g]));
            tomb_push(st, tomb_ldts, tomb_cap, op.pdel_ldt[g]);
        }
        for (uint32_t j = 0; j < op.row_count[g]; j++) {
            uint64_t o = op.row_base[g] + j;
            uint8_t kind = out.rkind[o];
            if (kind != BK_CLUSTERING) {
                lts(out.rdel_mfda[o]);
                lldt(ldt_long(out.rdel_ldt[o]));
                tomb_push(st, tomb_ldts, tomb_cap, out.rdel_ldt[o]);
                if (bk_is_boundary(kind)) {
                    lts(out.start_mfda[o]);
                    lldt(ldt_long(out.start_ldt[o]));
                    tomb_push(st, tomb_ldts, tomb_cap, out.start_ldt[o]);
                }
                continue;
            }
            atomicAdd(&sh_rows, 1ull);
            uint8_t of = out.flags[o];
            if (of & PF_LIVE_TS) {
                lts(out.live_ts[o]);
                if (out.live_ttl[o] != 0) {
                    lldt(out.live_let[o]);
                    atomicMin(&sh_minttl, (unsigned)out.live_ttl[o]);
                    atomicMax(&sh_maxttl, (unsigned)out.live_ttl[o]);
                } else lldt(NO_DELETION_TIME);
            }
            if (of & PF_ROW_DEL) {
                lts(out.rdel_mfda[o]);
                lldt(ldt_long(out.rdel_ldt[o]));
                tomb_push(st, tomb_ldts, tomb_cap, out.rdel_ldt[o]);
            }
            for (uint32_t c = 0; c < n_cols; c++) {
                uint64_t oc = o * n_cols + c;
                if (!(out.cell_flags[oc] & CELLF_PRESENT)) continue;
                atomicAdd(&sh_cells, 1ull);
                lts(out.cell_ts[oc]);
                uint32_t cldt = out.cell_ldt[oc];
                int32_t cttl = out.cell_ttl[oc];
                if (cldt != LDT_NONE_U32 && cttl == 0) { lldt(ldt_long(cldt)); tomb_push(st, tomb_ldts, tomb_cap, cldt); }
                else if (cttl != 0) {
                    lldt(ldt_long(cldt));
                    atomicMin(&sh_minttl, (unsigned)cttl);
                    atomicMax(&sh_maxttl, (unsigned)cttl);
                } else lldt(NO_DELETION_TIME);
            }
            if (of & PF_HAS_CPX) {
                // MetadataCollector.update(complexDeletion) + per-cell stats;
                // totalColumnsSet counts the COLUMN once when it has cells
                // (Rows.java:66-82)
                int64_t xdm = out.cpx_del_mfda[o];
                uint32_t xdl = out.cpx_del_ldt[o];
                if (!(xdm == INT64_MIN && xdl == LDT_NONE_U32)) {
                    lts(xdm);
                    lldt(ldt_long(xdl));
                    tomb_push(st, tomb_ldts, tomb_cap, xdl);
                }
                uint32_t nc2 = out.cpx_count[o];
                if (nc2) atomicAdd(&sh_cells, 1ull);
                uint64_t si = out.cpx_start[o];
                for (uint32_t e2 = 0; e2 < nc2; e2++) {
                    uint64_t xe = si + e2;
                    lts(out.cpx.ts[xe]);
                    uint32_t cldt = out.cpx.ldt[xe];
                    int32_t cttl = out.cpx.ttl[xe];
                    if (cldt != LDT_NONE_U32 && cttl == 0) { lldt(ldt_long(cldt)); tomb_push(st, tomb_ldts, tomb_cap, cldt); }
                    else if (cttl != 0) {
                        lldt(ldt_long(cldt));
                        atomicMin(&sh_minttl, (unsigned)cttl);
                        atomicMax(&sh_maxttl, (unsigned)cttl);
                    } else lldt(NO_DELETION_TIME);
                }
            }
        }
        atomicMin(&st->first_group, (unsigned long long)g);
        atomicMax(&st->last_group, (unsigned long long)g);
    }
    __syncthreads();
    if (threadIdx.x == 0 && sh_parts) {
        atomicAdd(&st->partitions_out, sh_parts);
        atomicAdd(&st->rows_out, sh_rows);
        atomicAdd(&st->total_cells, sh_cells);
        atomicMin(&st->min_ts_flip, sh_mints);
        atomicMax(&st->max_ts_flip, sh_maxts);
        atomicMin(&st->min_ldt_flip, sh_minldt);
        atomicMax(&st->max_ldt_flip, sh_maxldt);
        atomicMin(&st->min_ttl, sh_minttl);
        atomicMax(&st->max_ttl, sh_maxttl);
        if (sh_haspdel) atomicExch(&st->has_partition_deletions, 1ull);
    }
}

// ---------------------------------------------------------------------------
// garbage collect (nodetool garbagecollect / CompactionIterator.GarbageSkipper
// CompactionIterator.java:401-635): drop data shadowed by tombstone SOURCES.
// Two-phase: data and sources are merged separately (purge deferred), then
// a thread-per-partition state machine filters the merged data stream
// against the merged source stream, then k_purge_parts applies PurgeFunction.
// ---------------------------------------------------------------------------
__device__ inline bool dtp_sup(int64_t am, uint32_t al, int64_t bm, uint32_t bl) {
    return am > bm || (am == bm && ldt_long(al) > ldt_long(bl));
}
// cross-container position compare (same rules as pos_cmp)
__device__ inline int pos_cmp2(const UnfCols& A, uint64_t a, const UnfCols& B, uint64_t b,
                               const SchemaParams& sch) {
    if (sch.n_ck) {
        uint32_t na = A.ck_count[a], nb = B.ck_count[b];
        uint32_t mn = na < nb ? na : nb;
        for (uint32_t c = 0; c < mn; c++) {
            uint64_t ac = a * sch.n_ck + c, bc = b * sch.n_ck + c;
            uint64_t pa = A.ck[ac], pb = B.ck[bc];
            if (pa != pb) return pa < pb ? -1 : 1;
            if (sch.ck_w[c] < 0) {
                uint32_t la = A.ck_len[ac], lb = B.ck_len[bc];
                if (la > 8 || lb > 8) {
                    const uint8_t* a8 = (const uint8_t*)A.ck_addr[ac];
                    const uint8_t* b8 = (const uint8_t*)B.ck_addr[bc];
                    uint32_t n = la < lb ? la : lb;
                    for (uint32_t i = 8; i < n; i++)
                        if (a8[i] != b8[i]) return a8[i] < b8[i] ? -1 : 1;
                }
                if (la != lb) return la < lb ? -1 : 1;
            }
        }
        if (na != nb)
            return na < nb ? bk_cmp_to_clustering(A.rkind[a]) : -bk_cmp_to_clustering(B.rkind[b]);
    }
    return bk_comparison(A.rkind[a]) - bk_comparison(B.rkind[b]);
}
// Cells.reconcile(a, b) == b ? (the source cell wins)
__device__ inline bool gc_cell_b_wins(int64_t ats, uint32_t aldt, bool aexp, uint64_t av,
                                      uint32_t al, int64_t bts, uint32_t bldt, bool bexp,
                                      uint64_t bv, uint32_t bl) {
    if (ats != bts) return bts > ats;
    bool a_dt = aldt != LDT_NONE_U32, b_dt = bldt != LDT_NONE_U32;
    if (a_dt || b_dt) {
        if (a_dt != b_dt) return b_dt;
        bool a_tomb = !aexp, b_tomb = !bexp;
        if (a_tomb != b_tomb) return b_tomb;
        if (aldt != bldt) return ldt_long(bldt) > ldt_long(aldt);
        return cmp_values(av, al, bv, bl) < 0;
    }
    return cmp_values(av, al, bv, bl) < 0;
}

// pairwise CounterContext.merge relationship (CounterContext.merge over the
// ContextState walk; same lattice as ctx_compare in the oracle): 0 = merge
// returns LEFT's exact bytes (left superset or equal contexts — equality
// resolves left, matching the reference's left-identity-first check),
// 1 = returns RIGHT's bytes, 2 = fresh merged bytes.
__device__ inline int gc_ctx_relationship(const uint8_t* lp, uint32_t ll,
                                          const uint8_t* rp, uint32_t rl) {
    CtxSt L, R;
    L.init(lp, ll);
    R.init(rp, rl);
    bool lsup = true, rsup = true;
    while (L.has() && R.has()) {
        int c = ctx_idcmp(L.id(), R.id());
        if (c == 0) {
            int rel;
            int64_t lc = L.clock(), lk = L.count(), rc = R.clock(), rk = R.count();
            if (L.g || R.g) {
                if (L.g && R.g)
                    rel = lc == rc ? (lk > rk ? 1 : lk == rk ? 0 : -1) : (lc > rc ? 1 : -1);
                else
                    rel = L.g ? 1 : -1;
            } else if (L.l || R.l) {
                rel = (L.l && R.l) ? 2 : (L.l ? 1 : -1);
            } else if (lc == rc) {
                rel = lk > rk ? 1 : lk == rk ? 0 : -1;
            } else {
                rel = ctx_remote_better(lc, lk, rc, rk) ? 1 : -1;
            }
            if (rel == 2) lsup = rsup = false;
            else if (rel > 0) rsup = false;
            else if (rel < 0) lsup = false;
            L.next();
            R.next();
        } else if (c < 0) {
            rsup = false;
            L.next();
        } else {
            lsup = false;
            R.next();
        }
    }
    if (L.has()) rsup = false;
    if (R.has()) lsup = false;
    return lsup ? 0 : rsup ? 1 : 2;
}

// Cells.addNonShadowed identity decision for counter cells (resolveCounter,
// Cells.java:121-162): data cell dropped iff reconcile returns the source
// cell itself. Counter cells never expire, so tombstone == ldt set.
__device__ inline bool gc_ctr_b_wins(int64_t ats, uint32_t aldt, uint64_t av, uint32_t al,
                                     int64_t bts, uint32_t bldt, uint64_t bv, uint32_t bl) {
    bool at = aldt != LDT_NONE_U32, bt = bldt != LDT_NONE_U32;
    if (at || bt) {
        if (at != bt) return bt;  // tombstone beats any live counter
        return gc_cell_b_wins(ats, aldt, false, av, al, bts, bldt, false, bv, bl);
    }
    if (al == 0 || bl == 0) {  // non-empty wins (documented divergence)
        if ((al == 0) != (bl == 0)) return al == 0;
        return !(ats > bts);
    }
    int rel = gc_ctx_relationship((const uint8_t*)av, al, (const uint8_t*)bv, bl);
    return rel == 1 && (ats > bts ? ats : bts) == bts;
}

// move one complex cell within the shared cpx arena (row-local compaction)
__device__ inline void gc_cpx_move(UnfCols& u, uint64_t to, uint64_t from) {
    if (to == from) return;
    u.cpx.ts[to] = u.cpx.ts[from];
    u.cpx.ldt[to] = u.cpx.ldt[from];
    u.cpx.ttl[to] = u.cpx.ttl[from];
    u.cpx.flags[to] = u.cpx.flags[from];
    u.cpx.path_addr[to] = u.cpx.path_addr[from];
    u.cpx.path_len[to] = u.cpx.path_len[from];
    u.cpx.val_addr[to] = u.cpx.val_addr[from];
    u.cpx.val_len[to] = u.cpx.val_len[from];
}

// CellPath order (BytesType map keys): unsigned lex, shorter-first on ties
__device__ inline int gc_path_cmp(const UnfCols& a, uint64_t ae, const UnfCols& b, uint64_t be) {
    const uint8_t* pa = (const uint8_t*)a.cpx.path_addr[ae];
    const uint8_t* pb = (const uint8_t*)b.cpx.path_addr[be];
    uint32_t la = a.cpx.path_len[ae], lb = b.cpx.path_len[be];
    uint32_t n = la < lb ? la : lb;
    for (uint32_t i = 0; i < n; i++)
        if (pa[i] != pb[i]) return pa[i] < pb[i] ? -1 : 1;
    return la == lb ? 0 : (la < lb ? -1 : 1);
}

// per data partition: match to a source partition by decorated key, produce
// the filtered capacity (data rows + matched tomb rows for reissue headroom)
__global__ void k_gc_match(OutParts dp, uint64_t nd, OutParts tp, uint64_t nt,
                           int64_t* tomb_idx, uint64_t* cap) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= nd) return;
    cap[g] = dp.row_count[g];
    tomb_idx[g] = -1;
    if (!dp.keep[g] || nt == 0) return;
    int64_t tok = dp.token[g];
    const uint8_t* key = (const uint8_t*)dp.key_addr[g];
    uint32_t klen = dp.klen[g];
    uint64_t lo = 0, hi = nt;
    while (lo < hi) {  // lower_bound by (token, key bytes)
        uint64_t mid = (lo + hi) >> 1;
        int c = tp.token[mid] < tok ? -1 : tp.token[mid] > tok ? 1 : 0;
        if (c == 0) {
            const uint8_t* mk = (const uint8_t*)tp.key_addr[mid];
            uint32_t ml = tp.klen[mid];
            uint32_t n = ml < klen ? ml : klen;
            for (uint32_t i = 0; i < n && c == 0; i++)
                c = mk[i] < key[i] ? -1 : mk[i] > key[i] ? 1 : 0;
            if (c == 0) c = ml < klen ? -1 : ml > klen ? 1 : 0;
        }
        if (c < 0) lo = mid + 1;
        else hi = mid;
    }
    if (lo >= nt || tp.token[lo] != tok || tp.klen[lo] != klen) return;
    const uint8_t* mk = (const uint8_t*)tp.key_addr[lo];
    for (uint32_t i = 0; i < klen; i++)
        if (mk[i] != key[i]) return;
    tomb_idx[g] = (int64_t)lo;
    cap[g] = dp.row_count[g] + tp.row_count[lo];
}

// BTreeRow.filter(all, active, false): drop liveness/cells/deletion shadowed
// by `active`; returns row-non-empty. Operates on the OUTPUT arena row `o`.
__device__ inline bool gc_row_filter_active(UnfCols& out, uint64_t o, int64_t am, uint32_t al,
                                            const SchemaParams& sch) {
    bool active_live = am == INT64_MIN && al == LDT_NONE_U32;
    uint8_t f = out.flags[o];
    bool may = !active_live && !dtp_sup(out.rdel_mfda[o], out.rdel_ldt[o], am, al);
    bool any_cell = false;
    if (may) {
        if ((f & PF_LIVE_TS) && out.live_ts[o] <= am) {
            f &= ~PF_LIVE_TS;
            out.live_ts[o] = NO_TIMESTAMP;
            out.live_ttl[o] = 0;
            out.live_let[o] = NO_DELETION_TIME;
        }
        f &= ~PF_ROW_DEL;
        out.rdel_mfda[o] = INT64_MIN;
        out.rdel_ldt[o] = LDT_NONE_U32;
        for (uint32_t c = 0; c < sch.n_cols; c++) {
            uint64_t oc = o * sch.n_cols + c;
            if (!(out.cell_flags[oc] & CELLF_PRESENT)) continue;
            if (out.cell_ts[oc] <= am) out.cell_flags[oc] = 0;
            else any_cell = true;
        }
        if (sch.n_cpx && (f & PF_HAS_CPX)) {
            // ComplexColumnData.filter(all, activeDeletion, null, -)
            // (ComplexColumnData.java:188-210): drop a shadowed
            // complexDeletion, drop cells activeDeletion deletes
            if (dtp_sup(am, al, out.cpx_del_mfda[o], out.cpx_del_ldt[o])) {
                out.cpx_del_mfda[o] = INT64_MIN;
                out.cpx_del_ldt[o] = LDT_NONE_U32;
            }
            uint64_t s0 = out.cpx_start[o];
            uint32_t nc2 = out.cpx_count[o], w = 0;
            for (uint32_t e = 0; e < nc2; e++) {
                if (out.cpx.ts[s0 + e] <= am) continue;
                gc_cpx_move(out, s0 + w, s0 + e);
                w++;
            }
            out.cpx_count[o] = w;
            bool cpx_alive = w || !(out.cpx_del_mfda[o] == INT64_MIN &&
                                    out.cpx_del_ldt[o] == LDT_NONE_U32);
            if (!cpx_alive) f &= ~PF_HAS_CPX;
            else any_cell = true;
        }
    } else {
        for (uint32_t c = 0; c < sch.n_cols; c++)
            if (out.cell_flags[o * sch.n_cols + c] & CELLF_PRESENT) any_cell = true;
        if (sch.n_cpx && (f & PF_HAS_CPX)) any_cell = true;
    }
    bool keep = (f & (PF_LIVE_TS | PF_ROW_DEL)) || any_cell;
    out.flags[o] = keep ? (f | PF_HAS_ROW) : 0;
    return keep;
}

// garbageFilterRow on the output arena row `o` vs tomb arena row `t`
__device__ inline bool gc_filter_row_vs(UnfCols& out, uint64_t o, const UnfCols& tin, uint64_t t,
                                        int64_t am, uint32_t al, bool cell_level,
                                        const SchemaParams& sch) {
    int64_t tdm = (tin.flags[t] & PF_ROW_DEL) ? tin.rdel_mfda[t] : INT64_MIN;
    uint32_t tdl = (tin.flags[t] & PF_ROW_DEL) ? tin.rdel_ldt[t] : LDT_NONE_U32;
    int64_t dm = am;
    uint32_t dl = al;
    if (dtp_sup(tdm, tdl, dm, dl)) { dm = tdm; dl = tdl; }
    if (!cell_level) return gc_row_filter_active(out, o, dm, dl, sch);
    // Rows.removeShadowedCells
    uint8_t f = out.flags[o];
    if ((f & PF_LIVE_TS) && out.live_ts[o] <= dm) {
        f &= ~PF_LIVE_TS;
        out.live_ts[o] = NO_TIMESTAMP;
        out.live_ttl[o] = 0;
        out.live_let[o] = NO_DELETION_TIME;
    }
    if ((f & PF_ROW_DEL) && dtp_sup(dm, dl, out.rdel_mfda[o], out.rdel_ldt[o])) {
        f &= ~PF_ROW_DEL;
        out.rdel_mfda[o] = INT64_MIN;
        out.rdel_ldt[o] = LDT_NONE_U32;
    }
    bool any_cell = false;
    for (uint32_t c = 0; c < sch.n_cols; c++) {
        uint64_t oc = o * sch.n_cols + c, tc = t * sch.n_cols + c;
        uint8_t cf = out.cell_flags[oc];
        if (!(cf & CELLF_PRESENT)) continue;
        if (out.cell_ts[oc] <= dm) { out.cell_flags[oc] = 0; continue; }
        uint8_t tf = tin.cell_flags[tc];
        if (tf & CELLF_PRESENT) {
            bool bw = sch.counters
                          ? gc_ctr_b_wins(out.cell_ts[oc], out.cell_ldt[oc], out.val_addr[oc],
                                          out.val_len[oc], tin.cell_ts[tc], tin.cell_ldt[tc],
                                          tin.val_addr[tc], tin.val_len[tc])
                          : gc_cell_b_wins(out.cell_ts[oc], out.cell_ldt[oc],
                                           cf & CELLF_EXPIRING, out.val_addr[oc],
                                           out.val_len[oc], tin.cell_ts[tc], tin.cell_ldt[tc],
                                           tf & CELLF_EXPIRING, tin.val_addr[tc],
                                           tin.val_len[tc]);
            if (bw) {
                out.cell_flags[oc] = 0;
                continue;
            }
        }
        any_cell = true;
    }
    if (sch.n_cpx && (f & PF_HAS_CPX)) {
        // complex branch of Rows.removeShadowedCells (Rows.java:298-316):
        // data complexDeletion survives iff it supersedes
        // max(updateDt, deletion) and then raises the bar; then per-path
        // Cells.addNonShadowedComplex against the source's cells
        bool b_has = tin.flags[t] & PF_HAS_CPX;
        int64_t udm = b_has ? tin.cpx_del_mfda[t] : INT64_MIN;
        uint32_t udl = b_has ? tin.cpx_del_ldt[t] : LDT_NONE_U32;
        int64_t mm = dm;
        uint32_t ml = dl;
        if (dtp_sup(udm, udl, mm, ml)) { mm = udm; ml = udl; }
        bool keep_del = dtp_sup(out.cpx_del_mfda[o], out.cpx_del_ldt[o], mm, ml);
        if (keep_del) {
            mm = out.cpx_del_mfda[o];
            ml = out.cpx_del_ldt[o];
        } else {
            out.cpx_del_mfda[o] = INT64_MIN;
            out.cpx_del_ldt[o] = LDT_NONE_U32;
        }
        uint64_t s0 = out.cpx_start[o];
        uint32_t nc2 = out.cpx_count[o], w = 0;
        uint64_t tb0 = b_has ? tin.cpx_start[t] : 0;
        uint32_t tn = b_has ? tin.cpx_count[t] : 0, bi = 0;
        for (uint32_t e = 0; e < nc2; e++) {
            uint64_t xe = s0 + e;
            bool have_b = false;
            while (bi < tn) {
                int pcmp = gc_path_cmp(tin, tb0 + bi, out, xe);
                if (pcmp < 0) { bi++; continue; }
                have_b = pcmp == 0;
                break;
            }
            if (out.cpx.ts[xe] <= mm) continue;  // deletion.deletes(existing)
            if (have_b) {
                uint64_t te = tb0 + bi;
                if (gc_cell_b_wins(out.cpx.ts[xe], out.cpx.ldt[xe],
                                   out.cpx.flags[xe] & CELLF_EXPIRING, out.cpx.val_addr[xe],
                                   out.cpx.val_len[xe], tin.cpx.ts[te], tin.cpx.ldt[te],
                                   tin.cpx.flags[te] & CELLF_EXPIRING, tin.cpx.val_addr[te],
                                   tin.cpx.val_len[te]))
                    continue;
            }
            gc_cpx_move(out, s0 + w, xe);
            w++;
        }
        out.cpx_count[o] = w;
        if (!w && !keep_del) f &= ~PF_HAS_CPX;
        else any_cell = true;
    }
    bool keep = (f & (PF_LIVE_TS | PF_ROW_DEL)) || any_cell;
    out.flags[o] = keep ? (f | PF_HAS_ROW) : 0;
    return keep;
}

__device__ inline void gc_copy_unf(UnfCols& dst, uint64_t d, const UnfCols& src, uint64_t s,
                                   const SchemaParams& sch) {
    dst.rkind[d] = src.rkind[s];
    dst.flags[d] = src.flags[s];
    dst.live_ts[d] = src.live_ts[s];
    dst.live_ttl[d] = src.live_ttl[s];
    dst.live_let[d] = src.live_let[s];
    dst.rdel_mfda[d] = src.rdel_mfda[s];
    dst.rdel_ldt[d] = src.rdel_ldt[s];
    dst.start_mfda[d] = src.start_mfda[s];
    dst.start_ldt[d] = src.start_ldt[s];
    dst.ck_count[d] = src.ck_count[s];
    for (uint32_t c = 0; c < sch.n_ck; c++) {
        dst.ck[d * sch.n_ck + c] = src.ck[s * sch.n_ck + c];
        dst.ck_addr[d * sch.n_ck + c] = src.ck_addr[s * sch.n_ck + c];
        dst.ck_len[d * sch.n_ck + c] = src.ck_len[s * sch.n_ck + c];
    }
    for (uint32_t c = 0; c < sch.n_cols; c++) {
        uint64_t dc = d * sch.n_cols + c, sc = s * sch.n_cols + c;
        dst.cell_flags[dc] = src.cell_flags[sc];
        dst.cell_ts[dc] = src.cell_ts[sc];
        dst.cell_ldt[dc] = src.cell_ldt[sc];
        dst.cell_ttl[dc] = src.cell_ttl[sc];
        dst.val_addr[dc] = src.val_addr[sc];
        dst.val_len[dc] = src.val_len[sc];
    }
    if (sch.n_cpx) {
        // the cpx CELL arena is shared (aliased) between src and dst in the
        // gc path; only the per-row segment descriptors are copied
        dst.cpx_del_mfda[d] = src.cpx_del_mfda[s];
        dst.cpx_del_ldt[d] = src.cpx_del_ldt[s];
        dst.cpx_start[d] = src.cpx_start[s];
        dst.cpx_count[d] = src.cpx_count[s];
    }
}

__global__ void k_garbage_filter(OutParts dp, uint64_t nd, OutParts tp, UnfCols din,
                                 UnfCols tin, UnfCols dout, const int64_t* tomb_idx,
                                 const uint64_t* new_base, SchemaParams sch, int cell_level) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= nd) return;
    uint64_t ob = new_base[g];
    if (!dp.keep[g]) { dp.row_base[g] = ob; dp.row_count[g] = 0; return; }
    int64_t tg = tomb_idx[g];
    uint64_t db = dp.row_base[g];
    uint32_t dn = dp.row_count[g];
    if (tg < 0) {  // no shadow source for this key: copy through
        for (uint32_t j = 0; j < dn; j++) gc_copy_unf(dout, ob + j, din, db + j, sch);
        dp.row_base[g] = ob;
        return;
    }
    uint64_t tb = tp.row_base[tg];
    uint32_t tn = tp.row_count[tg];
    // partition deletion: keep only if it supersedes the source's
    int64_t pdm = tp.pdel_mfda[tg];
    uint32_t pdl = tp.pdel_ldt[tg];
    if (!dtp_sup(dp.pdel_mfda[g], dp.pdel_ldt[g], pdm, pdl)) {
        dp.pdel_mfda[g] = INT64_MIN;
        dp.pdel_ldt[g] = LDT_NONE_U32;
    }
    int64_t act_m = pdm;     // activeDeletionTime (starts at source partition deletion)
    uint32_t act_l = pdl;
    // static row vs source static row
    if (sch.n_static && (dp.st.flags[g] & PF_HAS_ROW)) {
        // reuse the row machinery on the partition-level static SoA via a
        // small shim: treat static columns as "cells" of a 1-row UnfCols view
        // (fields are laid out identically), simplest done inline:
        int64_t sdm = act_m;
        uint32_t sdl = act_l;
        if (tp.st.flags[tg] & PF_ROW_DEL)
            if (dtp_sup(tp.st.rdel_mfda[tg], tp.st.rdel_ldt[tg], sdm, sdl)) {
                sdm = tp.st.rdel_mfda[tg];
                sdl = tp.st.rdel_ldt[tg];
            }
        uint8_t f = dp.st.flags[g];
        bool active_live = sdm == INT64_MIN && sdl == LDT_NONE_U32;
        bool may = !active_live && !dtp_sup(dp.st.rdel_mfda[g], dp.st.rdel_ldt[g], sdm, sdl);
        bool any_cell = false;
        if (cell_level || may) {
            int64_t lim = cell_level ? sdm : (may ? sdm : INT64_MIN);
            if ((f & PF_LIVE_TS) && dp.st.live_ts[g] <= lim) {
                f &= ~PF_LIVE_TS;
                dp.st.live_ts[g] = NO_TIMESTAMP;
            }
            if (!cell_level) {
                f &= ~PF_ROW_DEL;
                dp.st.rdel_mfda[g] = INT64_MIN;
                dp.st.rdel_ldt[g] = LDT_NONE_U32;
            } else if ((f & PF_ROW_DEL) && dtp_sup(sdm, sdl, dp.st.rdel_mfda[g], dp.st.rdel_ldt[g])) {
                f &= ~PF_ROW_DEL;
                dp.st.rdel_mfda[g] = INT64_MIN;
                dp.st.rdel_ldt[g] = LDT_NONE_U32;
            }
            for (uint32_t c = 0; c < sch.n_static; c++) {
                uint64_t oc = g * sch.n_static + c, tc = (uint64_t)tg * sch.n_static + c;
                uint8_t cf = dp.st.cell_flags[oc];
                if (!(cf & CELLF_PRESENT)) continue;
                if (dp.st.cell_ts[oc] <= lim) { dp.st.cell_flags[oc] = 0; continue; }
                uint8_t tf = tp.st.cell_flags[tc];
                if (cell_level && (tf & CELLF_PRESENT) &&
                    gc_cell_b_wins(dp.st.cell_ts[oc], dp.st.cell_ldt[oc], cf & CELLF_EXPIRING,
                                   dp.st.val_addr[oc], dp.st.val_len[oc], tp.st.cell_ts[tc],
                                   tp.st.cell_ldt[tc], tf & CELLF_EXPIRING, tp.st.val_addr[tc],
                                   tp.st.val_len[tc])) {
                    dp.st.cell_flags[oc] = 0;
                    continue;
                }
                any_cell = true;
            }
            dp.st.flags[g] = ((f & (PF_LIVE_TS | PF_ROW_DEL)) || any_cell) ? (f | PF_HAS_ROW) : 0;
        }
    }
    // ---- two-cursor stream filter (GarbageSkippingUnfilteredRowIterator) ----
    int64_t tomb_open_m = INT64_MIN, data_open_m = INT64_MIN, open_m = INT64_MIN;
    uint32_t tomb_open_l = LDT_NONE_U32, data_open_l = LDT_NONE_U32, open_l = LDT_NONE_U32;
    uint32_t di = 0, ti = 0, oc = 0;
    auto upd_open = [&](const UnfCols& in, uint64_t o, int64_t* m, uint32_t* l) {
        uint8_t kk = in.rkind[o];
        if (bk_is_open(kk)) {
            *m = bk_is_boundary(kk) ? in.start_mfda[o] : in.rdel_mfda[o];
            *l = bk_is_boundary(kk) ? in.start_ldt[o] : in.rdel_ldt[o];
        } else {
            *m = INT64_MIN;
            *l = LDT_NONE_U32;
        }
    };
    while (di < dn) {
        uint64_t d = db + di;
        int cmp = ti >= tn ? -1 : pos_cmp2(din, d, tin, tb + ti, sch);
        bool have = false;
        uint64_t slot = ob + oc;
        auto process_data_marker = [&]() {
            upd_open(din, d, &data_open_m, &data_open_l);
            bool before = open_m == INT64_MIN && open_l == LDT_NONE_U32;
            bool after = !dtp_sup(data_open_m, data_open_l, act_m, act_l);
            uint8_t kk = din.rkind[d];
            if (!before && !after) {
                gc_copy_unf(dout, slot, din, d, sch);
                have = true;
            } else if (!before && after) {
                gc_copy_unf(dout, slot, din, d, sch);
                dout.rkind[slot] = kk == BK_EXCL_END_INCL_START ? BK_EXCL_END
                                   : kk == BK_INCL_END_EXCL_START ? BK_INCL_END : kk;
                dout.start_mfda[slot] = INT64_MIN;
                dout.start_ldt[slot] = LDT_NONE_U32;
                have = true;
            } else if (before && !after) {
                gc_copy_unf(dout, slot, din, d, sch);
                dout.rkind[slot] = kk == BK_EXCL_END_INCL_START ? BK_INCL_START
                                   : kk == BK_INCL_END_EXCL_START ? BK_EXCL_START : kk;
                // open-only bound: deletion = open_dt of the original
                dout.rdel_mfda[slot] = bk_is_boundary(kk) ? din.start_mfda[d] : din.rdel_mfda[d];
                dout.rdel_ldt[slot] = bk_is_boundary(kk) ? din.start_ldt[d] : din.rdel_ldt[d];
                dout.start_mfda[slot] = INT64_MIN;
                dout.start_ldt[slot] = LDT_NONE_U32;
                have = true;
            }
        };
        if (cmp < 0) {
            if (din.rkind[d] == BK_CLUSTERING) {
                gc_copy_unf(dout, slot, din, d, sch);
                have = gc_row_filter_active(dout, slot, act_m, act_l, sch);
            } else {
                process_data_marker();
            }
        } else if (cmp == 0) {
            uint64_t t = tb + ti;
            if (din.rkind[d] == BK_CLUSTERING) {
                gc_copy_unf(dout, slot, din, d, sch);
                have = gc_filter_row_vs(dout, slot, tin, t, act_m, act_l, cell_level != 0, sch);
            } else {
                upd_open(tin, t, &tomb_open_m, &tomb_open_l);
                act_m = pdm;
                act_l = pdl;
                if (dtp_sup(tomb_open_m, tomb_open_l, act_m, act_l)) { act_m = tomb_open_m; act_l = tomb_open_l; }
                process_data_marker();
            }
        } else {
            uint64_t t = tb + ti;
            if (tin.rkind[t] != BK_CLUSTERING) {
                upd_open(tin, t, &tomb_open_m, &tomb_open_l);
                act_m = pdm;
                act_l = pdl;
                if (dtp_sup(tomb_open_m, tomb_open_l, act_m, act_l)) { act_m = tomb_open_m; act_l = tomb_open_l; }
                bool before = open_m == INT64_MIN && open_l == LDT_NONE_U32;
                bool after = !dtp_sup(data_open_m, data_open_l, act_m, act_l);
                if (before && !after) {
                    // reopen the data range at the inverted source close bound
                    uint8_t tk = tin.rkind[t];
                    uint8_t ck2 = tk == BK_EXCL_END_INCL_START ? BK_EXCL_END
                                  : tk == BK_INCL_END_EXCL_START ? BK_INCL_END : tk;
                    uint8_t inv = ck2 == BK_INCL_END ? BK_EXCL_START
                                  : ck2 == BK_EXCL_END ? BK_INCL_START : ck2;
                    gc_copy_unf(dout, slot, tin, t, sch);  // clustering position
                    dout.flags[slot] = 0;
                    dout.rkind[slot] = inv;
                    dout.rdel_mfda[slot] = data_open_m;
                    dout.rdel_ldt[slot] = data_open_l;
                    dout.start_mfda[slot] = INT64_MIN;
                    dout.start_ldt[slot] = LDT_NONE_U32;
                    dout.live_ts[slot] = NO_TIMESTAMP;
                    dout.live_ttl[slot] = 0;
                    dout.live_let[slot] = NO_DELETION_TIME;
                    for (uint32_t c = 0; c < sch.n_cols; c++)
                        dout.cell_flags[slot * sch.n_cols + c] = 0;
                    have = true;
                }
            }
        }
        if (have && dout.rkind[slot] != BK_CLUSTERING)
            upd_open(dout, slot, &open_m, &open_l);
        if (cmp <= 0) di++;
        if (cmp >= 0) ti++;
        if (have) oc++;
    }
    dp.row_base[g] = ob;
    dp.row_count[g] = oc;
}

// PurgeFunction pass over an OutParts arena (the garbage-collect path defers
// purge until after the filter; mirrors oracle purge_partition). Compacts
// kept unfiltereds in place and refreshes keep/first/last bookkeeping.
__global__ void k_purge_parts(OutParts op, UnfCols out, uint64_t n, SchemaParams sch,
                              PurgeParams2 pp, OutStats* st) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n) return;
    if (!op.keep[g]) return;
    int64_t token = op.token[g];
    if (pp.ov_has_bloom) {
        uint64_t h[2];
        murmur3_128((const uint8_t*)op.key_addr[g], op.klen[g], 0, h);
        pp.kb = (int64_t)h[1];
        pp.ki = (int64_t)h[0];
    }
    auto purge_dt = [&](int64_t m, uint32_t l) {
        return m != INT64_MIN && should_purge2(pp, token, m, ldt_long(l));
    };
    // partition deletion
    if (!(op.pdel_mfda[g] == INT64_MIN && op.pdel_ldt[g] == LDT_NONE_U32) &&
        should_purge2(pp, token, op.pdel_mfda[g], ldt_long(op.pdel_ldt[g]))) {
        op.pdel_mfda[g] = INT64_MIN;
        op.pdel_ldt[g] = LDT_NONE_U32;
    }
    // static row
    bool static_kept = false;
    if (sch.n_static && (op.st.flags[g] & PF_HAS_ROW)) {
        uint8_t f = op.st.flags[g];
        if (f & PF_LIVE_TS) {
            int32_t ttl = op.st.live_ttl[g];
            int64_t let = op.st.live_let[g];
            bool is_live = ttl == INT32_MAX ? false : (ttl != 0 ? pp.now_sec < let : true);
            if (!is_live && should_purge2(pp, token, op.st.live_ts[g], let)) {
                f &= ~PF_LIVE_TS;
                op.st.live_ts[g] = NO_TIMESTAMP;
                op.st.live_ttl[g] = 0;
                op.st.live_let[g] = NO_DELETION_TIME;
            }
        }
        if ((f & PF_ROW_DEL) && purge_dt(op.st.rdel_mfda[g], op.st.rdel_ldt[g])) {
            f &= ~PF_ROW_DEL;
            op.st.rdel_mfda[g] = INT64_MIN;
            op.st.rdel_ldt[g] = LDT_NONE_U32;
        }
        bool any_cell = false;
        for (uint32_t c = 0; c < sch.n_static; c++) {
            uint64_t oc = g * sch.n_static + c;
            uint8_t cf = op.st.cell_flags[oc];
            if (!(cf & CELLF_PRESENT)) continue;
            int64_t cts = op.st.cell_ts[oc];
            uint32_t cldt = op.st.cell_ldt[oc];
            int32_t cttl = op.st.cell_ttl[oc];
            bool live_cell = cldt == LDT_NONE_U32 || (cttl != 0 && pp.now_sec < ldt_long(cldt));
            if (!live_cell) {
                if (should_purge2(pp, token, cts, ldt_long(cldt))) { op.st.cell_flags[oc] = 0; continue; }
                if (cttl != 0) {
                    int64_t nldt = ldt_long(cldt) - cttl;
                    if (should_purge2(pp, token, cts, nldt)) { op.st.cell_flags[oc] = 0; continue; }
                    op.st.cell_ldt[oc] = ldt_u32(nldt);
                    op.st.cell_ttl[oc] = 0;
                    op.st.cell_flags[oc] = CELLF_PRESENT;
                    op.st.val_len[oc] = 0;
                }
            }
            any_cell = true;
        }
        op.st.flags[g] = ((f & (PF_LIVE_TS | PF_ROW_DEL)) || any_cell) ? (f | PF_HAS_ROW) : 0;
        static_kept = (op.st.flags[g] & PF_HAS_ROW) != 0;
    }
    // unfiltereds
    uint64_t rb = op.row_base[g];
    uint32_t nrows = op.row_count[g];
    uint32_t w = 0;
    for (uint32_t j = 0; j < nrows; j++) {
        uint64_t o = rb + j;
        uint8_t kk = out.rkind[o];
        bool keep_u = false;
        if (kk != BK_CLUSTERING) {
            if (bk_is_boundary(kk)) {
                bool pclose = purge_dt(out.rdel_mfda[o], out.rdel_ldt[o]);
                bool popen = purge_dt(out.start_mfda[o], out.start_ldt[o]);
                if (pclose && popen) keep_u = false;
                else if (pclose) {
                    out.rkind[o] = kk == BK_EXCL_END_INCL_START ? BK_INCL_START : BK_EXCL_START;
                    out.rdel_mfda[o] = out.start_mfda[o];
                    out.rdel_ldt[o] = out.start_ldt[o];
                    out.start_mfda[o] = INT64_MIN;
                    out.start_ldt[o] = LDT_NONE_U32;
                    keep_u = true;
                } else if (popen) {
                    out.rkind[o] = kk == BK_EXCL_END_INCL_START ? BK_EXCL_END : BK_INCL_END;
                    out.start_mfda[o] = INT64_MIN;
                    out.start_ldt[o] = LDT_NONE_U32;
                    keep_u = true;
                } else keep_u = true;
            } else {
                keep_u = !purge_dt(out.rdel_mfda[o], out.rdel_ldt[o]);
            }
        } else {
            uint8_t f = out.flags[o];
            if (f & PF_LIVE_TS) {
                int32_t ttl = out.live_ttl[o];
                int64_t let = out.live_let[o];
                bool is_live = ttl == INT32_MAX ? false : (ttl != 0 ? pp.now_sec < let : true);
                if (!is_live && should_purge2(pp, token, out.live_ts[o], let)) {
                    f &= ~PF_LIVE_TS;
                    out.live_ts[o] = NO_TIMESTAMP;
                    out.live_ttl[o] = 0;
                    out.live_let[o] = NO_DELETION_TIME;
                }
            }
            if ((f & PF_ROW_DEL) && purge_dt(out.rdel_mfda[o], out.rdel_ldt[o])) {
                f &= ~PF_ROW_DEL;
                out.rdel_mfda[o] = INT64_MIN;
                out.rdel_ldt[o] = LDT_NONE_U32;
            }
            bool any_cell = false;
            for (uint32_t c = 0; c < sch.n_cols; c++) {
                uint64_t oc = o * sch.n_cols + c;
                uint8_t cf = out.cell_flags[oc];
                if (!(cf & CELLF_PRESENT)) continue;
                int64_t cts = out.cell_ts[oc];
                uint32_t cldt = out.cell_ldt[oc];
                int32_t cttl = out.cell_ttl[oc];
                bool live_cell = cldt == LDT_NONE_U32 || (cttl != 0 && pp.now_sec < ldt_long(cldt));
                if (!live_cell) {
                    if (should_purge2(pp, token, cts, ldt_long(cldt))) { out.cell_flags[oc] = 0; continue; }
                    if (cttl != 0) {
                        int64_t nldt = ldt_long(cldt) - cttl;
                        if (should_purge2(pp, token, cts, nldt)) { out.cell_flags[oc] = 0; continue; }
                        out.cell_ldt[oc] = ldt_u32(nldt);
                        out.cell_ttl[oc] = 0;
                        out.cell_flags[oc] = CELLF_PRESENT;
                        out.val_len[oc] = 0;
                    }
                }
                any_cell = true;
            }
            if (sch.n_cpx && (f & PF_HAS_CPX)) {
                // ComplexColumnData.purge (ComplexColumnData.java:212-216),
                // deferred-purge variant over the shared cpx arena
                if (out.cpx_del_mfda[o] != INT64_MIN &&
                    should_purge2(pp, token, out.cpx_del_mfda[o], ldt_long(out.cpx_del_ldt[o]))) {
                    out.cpx_del_mfda[o] = INT64_MIN;
                    out.cpx_del_ldt[o] = LDT_NONE_U32;
                }
                uint64_t s0 = out.cpx_start[o];
                uint32_t nc2 = out.cpx_count[o], wx = 0;
                for (uint32_t e = 0; e < nc2; e++) {
                    uint64_t xe = s0 + e;
                    int64_t cts = out.cpx.ts[xe];
                    uint32_t cldt = out.cpx.ldt[xe];
                    int32_t cttl = out.cpx.ttl[xe];
                    bool live_cell =
                        cldt == LDT_NONE_U32 || (cttl != 0 && pp.now_sec < ldt_long(cldt));
                    if (!live_cell) {
                        if (should_purge2(pp, token, cts, ldt_long(cldt))) continue;
                        if (cttl != 0) {
                            int64_t nldt = ldt_long(cldt) - cttl;
                            if (should_purge2(pp, token, cts, nldt)) continue;
                            out.cpx.ldt[xe] = ldt_u32(nldt);
                            out.cpx.ttl[xe] = 0;
                            out.cpx.flags[xe] = CELLF_PRESENT;  // expired -> tombstone
                            out.cpx.val_len[xe] = 0;
                        }
                    }
                    gc_cpx_move(out, s0 + wx, xe);
                    wx++;
                }
                out.cpx_count[o] = wx;
                if (!wx && out.cpx_del_mfda[o] == INT64_MIN &&
                    out.cpx_del_ldt[o] == LDT_NONE_U32)
                    f &= ~PF_HAS_CPX;
                else
                    any_cell = true;
            }
            if (!(f & (PF_LIVE_TS | PF_ROW_DEL)) && !any_cell) f = 0;
            else if (pp.enforce_strict_liveness && !(f & PF_LIVE_TS) && !(f & PF_ROW_DEL)) f = 0;
            out.flags[o] = f ? (f | PF_HAS_ROW) : 0;
            keep_u = (out.flags[o] & PF_HAS_ROW) != 0;
        }
        if (keep_u) {
            if (w != j) gc_copy_unf(out, rb + w, out, o, sch);
            w++;
        }
    }
    op.row_count[g] = w;
    bool pdel_live = op.pdel_mfda[g] == INT64_MIN && op.pdel_ldt[g] == LDT_NONE_U32;
    op.keep[g] = (!pdel_live || w > 0 || static_kept) ? 1 : 0;
    if (op.keep[g]) {
        atomicMin(&st->first_group, (unsigned long long)g);
        atomicMax(&st->last_group, (unsigned long long)g);
    }
}

// ---------------------------------------------------------------------------
// flush path (Memtable -> sstable): host hands unsorted unique-key rows in
// flat arenas; token+sort on device, then the shared writer kernels
// ---------------------------------------------------------------------------
struct FlushParams {
    const uint8_t* keys;      // concatenated key bytes
    const uint64_t* key_off;  // n+1 offsets into keys
    const int64_t* ts;
    const uint8_t* vals;      // concatenated value bytes
    const uint64_t* val_off;  // n+1 offsets (del rows: off[i+1]==off[i])
    const uint32_t* del_ldt;  // LDT_NONE_U32 = live row
    uint64_t n;
};

__global__ void k_flush_recs(FlushParams fp, MRec* recs) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= fp.n) return;
    const uint8_t* key = fp.keys + fp.key_off[i];
    uint32_t klen = (uint32_t)(fp.key_off[i + 1] - fp.key_off[i]);
    int64_t tok = murmur3_token(key, klen);
    uint64_t pfx = 0;
    uint32_t pb = klen < 8 ? klen : 8;
    for (uint32_t b = 0; b < pb; b++) pfx |= (uint64_t)key[b] << (8 * (7 - b));
    recs[i] = MRec{(uint64_t)tok ^ 0x8000000000000000ULL, pfx, (uint32_t)i, 0, (uint16_t)klen};
}

// uniqueness check: any adjacent equal (tok,pfx,klen) pair with equal bytes
__global__ void k_flush_dupcheck(FlushParams fp, const MRec* recs, unsigned long long* error) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i == 0 || i >= fp.n) return;
    const MRec &a = recs[i - 1], &b = recs[i];
    if (a.tok != b.tok || a.pfx != b.pfx || a.klen != b.klen) return;
    const uint8_t* ka = fp.keys + fp.key_off[a.idx];
    const uint8_t* kb = fp.keys + fp.key_off[b.idx];
    for (uint32_t x = 0; x < a.klen; x++)
        if (ka[x] != kb[x]) return;
    atomicExch(error, 40ull);
}

__global__ void k_flush_fill(FlushParams fp, const MRec* sorted, OutParts op, UnfCols out) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= fp.n) return;
    uint32_t r = sorted[i].idx;
    op.keypfx[i] = sorted[i].pfx;
    op.token[i] = (int64_t)(sorted[i].tok ^ 0x8000000000000000ULL);
    op.key_addr[i] = (uint64_t)(fp.keys + fp.key_off[r]);
    op.klen[i] = sorted[i].klen;
    op.pdel_mfda[i] = INT64_MIN;
    op.pdel_ldt[i] = LDT_NONE_U32;
    op.row_base[i] = i;
    op.row_count[i] = 1;
    op.keep[i] = 1;
    uint64_t o = i;
    out.rkind[o] = BK_CLUSTERING;
    out.ck_count[o] = 0;
    out.start_mfda[o] = INT64_MIN;
    out.start_ldt[o] = LDT_NONE_U32;
    out.live_ttl[o] = 0;
    out.live_let[o] = NO_DELETION_TIME;
    bool tomb = fp.del_ldt[r] != LDT_NONE_U32;
    if (tomb) {
        out.flags[o] = PF_HAS_ROW | PF_ROW_DEL;
        out.live_ts[o] = NO_TIMESTAMP;
        out.rdel_mfda[o] = fp.ts[r];
        out.rdel_ldt[o] = fp.del_ldt[r];
        out.cell_flags[o] = 0;
    } else {
        out.flags[o] = PF_HAS_ROW | PF_LIVE_TS;
        out.live_ts[o] = fp.ts[r];
        out.rdel_mfda[o] = INT64_MIN;
        out.rdel_ldt[o] = LDT_NONE_U32;
        out.cell_flags[o] = CELLF_PRESENT | CELLF_HAS_VALUE;
        out.cell_ts[o] = fp.ts[r];
        out.cell_ldt[o] = LDT_NONE_U32;
        out.cell_ttl[o] = 0;
        out.val_addr[o] = (uint64_t)(fp.vals + fp.val_off[r]);
        out.val_len[o] = (uint32_t)(fp.val_off[r + 1] - fp.val_off[r]);
    }
}

// ---------------------------------------------------------------------------
// verify epilogue (Verifier.java semantics: key order + bloom recheck)
// ---------------------------------------------------------------------------
// adjacent DecoratedKey order: every partition must sort strictly after its
// predecessor (Verifier.java "out of order" check)
__global__ void k_verify_order(const MRec* recs, uint64_t n, KeyLut lut,
                               unsigned long long* error) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i == 0 || i >= n) return;
    if (!mrec_less(lut, recs[i - 1], recs[i])) atomicExch(error, 30ull);
}

// recompute bloom bits from the parsed keys (FilterComponent recheck)
__global__ void k_verify_bloom(ParsedCols pc, const MRec* recs, uint64_t n,
                               uint32_t* bloom_bits, uint64_t bloom_bitlen, int32_t bloom_k) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t h[2];
    murmur3_128((const uint8_t*)pc.key_addr[i], recs[i].klen, 0, h);
    int64_t base = (int64_t)h[1], inc = (int64_t)h[0];
    for (int j = 0; j < bloom_k; j++) {
        int64_t m = base % (int64_t)bloom_bitlen;
        uint64_t idx = (uint64_t)((m ^ (m >> 63)) - (m >> 63));
        atomicOr(&bloom_bits[idx >> 5], 1u << (idx & 31));
        base += inc;
    }
}

// ---------------------------------------------------------------------------
// generator (general): thread per partition fills rows (+ markers) mirroring
// oracle/src/gen.h; values generated per live row into the values arena.
// ---------------------------------------------------------------------------
struct GenParams2 {
    uint64_t seed, universe, stride, rows;
    uint32_t sst;
    uint32_t value_len, value_repeat_pct, tombstone_pct, partition_del_pct;
    uint32_t clustering_rows, range_tomb_pct;
    uint32_t key_len;   // 8 (default) .. 255; bytes 8.. are gen2_key_salt(id, j)
    uint32_t ck_text;   // clustering values as UTF8 strings (oracle gen_ck_bytes)
    uint32_t ck_cols;   // 0/1 = one clustering column; 2 = (bigint, bigint)
    uint32_t static_pct;  // percent of wide partitions with a static row ("s0" blob)
    uint32_t n_value_cols;    // regular columns val0..valN-1 (1..63)
    uint32_t col_missing_pct; // P(cell absent) per live row and column
    uint32_t ttl_pct;         // P(live row written expiring) — oracle gen_row_expiring
    uint32_t complex_pct;     // P(live row carries 'zm' map cells) — gen_has_complex
    uint32_t complex_del_pct; // P(those rows also carry a complexDeletion)
    // counter mode (oracle gen.h gen_ctr_*): one CounterColumnType column;
    // ctr_pool = the 8 CounterIds in SORTED byte order, ctr_pool_idx their
    // original indices (role = idx % 3: global/local/remote)
    uint32_t counter;
    uint8_t ctr_pool[8][16];
    uint8_t ctr_pool_idx[8];
    int64_t base_ts, base_ldt;
};

// shared generator contract (oracle/src/gen.h): key = 8-byte BE id, then
// salt bytes for key_len > 8
// text clustering value (oracle gen_ck_bytes contract): 8 zero-padded
// decimal digits of the numeric position; rows append (pos/16 %% 3) 'x' bytes
__device__ inline uint32_t gen2_ck_text(int64_t ckval, bool is_row, uint8_t* out) {
    uint64_t v = (uint64_t)ckval;
    for (int i = 7; i >= 0; i--) { out[i] = (uint8_t)('0' + v % 10); v /= 10; }
    uint32_t len = 8;
    if (is_row) {
        uint32_t sfx = (uint32_t)(((uint64_t)ckval / 16) % 3);
        for (uint32_t i = 0; i < sfx; i++) out[len + i] = 'x';
        len += sfx;
    }
    return len;
}
__device__ __host__ inline uint8_t gen2_key_salt(uint64_t id, uint32_t j) {
    return (uint8_t)splitmix64(id ^ (0xC0FFEE5EEDULL + j));
}

__global__ void k_gen_recs2(GenParams2 gp, MRec* recs, uint64_t* ids, uint8_t* keys) {
    uint64_t j = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (j >= gp.rows) return;
    uint64_t id = feistel_perm(gp.seed, gp.universe, (gp.sst * gp.stride + j) % gp.universe);
    ids[j] = id;
    uint8_t* key = keys + j * gp.key_len;
    for (int b = 0; b < 8; b++) key[b] = (uint8_t)(id >> (8 * (7 - b)));
    for (uint32_t b = 8; b < gp.key_len; b++) key[b] = gen2_key_salt(id, b);
    int64_t tok = murmur3_token(key, gp.key_len);
    recs[j] = MRec{(uint64_t)tok ^ 0x8000000000000000ULL, id, (uint32_t)j, 0, (uint16_t)gp.key_len};
}

__device__ inline uint32_t gen2_ldt(const GenParams2& gp, uint64_t key_id, uint64_t salt) {
    return (uint32_t)(gp.base_ldt + (int64_t)(splitmix64(key_id ^ salt) % 1000));
}

// complex-column derivations (oracle gen.h: gen_has_complex/gen_cpx_*):
// candidate paths from a 40-value space, dedup+sorted per row
__device__ inline bool gen2_has_cpx(const GenParams2& gp, uint64_t id, uint32_t j) {
    if (gp.complex_pct == 0) return false;
    return splitmix64(gp.seed ^ 0xC0113C71ULL ^ id ^ ((uint64_t)gp.sst << 32) ^
                      (uint64_t)(j + 5) * 157) % 100 < gp.complex_pct;
}
__device__ inline bool gen2_has_cpx_del(const GenParams2& gp, uint64_t id, uint32_t j) {
    if (gp.complex_del_pct == 0) return false;
    return splitmix64(gp.seed ^ 0xCDE1CDE1ULL ^ id ^ ((uint64_t)gp.sst << 32) ^
                      (uint64_t)(j + 2) * 211) % 100 < gp.complex_del_pct;
}
__device__ inline uint32_t gen2_cpx_paths(const GenParams2& gp, uint64_t id, uint32_t j,
                                          uint32_t* pv) {
    uint32_t nc = 1 + (uint32_t)(splitmix64(id ^ 0xCE11C07ULL ^ (uint64_t)(j + 1) * 19) % 4);
    for (uint32_t e = 0; e < nc; e++)
        pv[e] = (uint32_t)(splitmix64(id ^ 0x9A7B9A7BULL ^ ((uint64_t)gp.sst << 24) ^
                                      (uint64_t)(j + 1) * 23 ^ (uint64_t)(e + 1) * 71) % 40);
    // insertion sort + unique over <=4 values
    for (uint32_t a = 1; a < nc; a++) {
        uint32_t v = pv[a];
        int b = (int)a - 1;
        while (b >= 0 && pv[b] > v) { pv[b + 1] = pv[b]; b--; }
        pv[b + 1] = v;
    }
    uint32_t w = nc ? 1 : 0;
    for (uint32_t a = 1; a < nc; a++)
        if (pv[a] != pv[a - 1]) pv[w++] = pv[a];
    return w;
}

// per-partition unfiltered count (upper bound used for layout: exact)
__global__ void k_gen_count(GenParams2 gp, const MRec* sorted, const uint64_t* ids, uint64_t n,
                            OutParts op, uint64_t* prow_count, const uint8_t* keys,
                            uint64_t* cpx_count = nullptr) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t id = ids[sorted[i].idx];
    if (cpx_count) {
        uint64_t xc = 0;
        uint32_t pv[4];
        uint32_t nrows = gp.clustering_rows ? gp.clustering_rows : 1;
        bool pdel = gp.clustering_rows == 0 && gp.partition_del_pct &&
                    (splitmix64(gp.seed ^ 0xFEEDULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 < gp.partition_del_pct);
        if (!pdel)
            for (uint32_t j = 0; j < nrows; j++) {
                bool tomb = gp.tombstone_pct &&
                            (splitmix64(gp.seed ^ 0xDEADULL ^ id ^ ((uint64_t)gp.sst << 32) ^
                                        (gp.clustering_rows ? (uint64_t)(j + 7) * 131 : 0)) % 100 <
                             gp.tombstone_pct);
                if (!tomb && gen2_has_cpx(gp, id, j)) xc += gen2_cpx_paths(gp, id, j, pv);
            }
        cpx_count[i] = xc;
    }
    uint32_t cnt;
    if (gp.clustering_rows == 0) {
        bool pdel = gp.partition_del_pct && (splitmix64(gp.seed ^ 0xFEEDULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 < gp.partition_del_pct);
        cnt = pdel ? 0 : 1;
    } else {
        cnt = gp.clustering_rows;
        if (gp.range_tomb_pct && (splitmix64(gp.seed ^ 0xBEEFULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 < gp.range_tomb_pct)) {
            uint64_t r = splitmix64(gp.seed ^ 0xB00BULL ^ id ^ ((uint64_t)gp.sst << 32));
            uint32_t a = (uint32_t)(r % gp.clustering_rows);
            // 1-col: bounds sit between rows, so both are emitted only when a
            // row follows the open position; 2-col prefix bounds always open
            // before their ck0 group's first row
            if (gp.ck_cols == 2 || a + 1 < gp.clustering_rows) cnt += 2;
        }
    }
    prow_count[i] = cnt;
    op.keypfx[i] = sorted[i].pfx;
    op.token[i] = (int64_t)(sorted[i].tok ^ 0x8000000000000000ULL);
    op.key_addr[i] = (uint64_t)(keys + (uint64_t)sorted[i].idx * gp.key_len);
    op.klen[i] = (uint16_t)gp.key_len;
    op.keep[i] = 1;
}

__global__ void k_gen_fill2(GenParams2 gp, const MRec* sorted, const uint64_t* ids, uint64_t n,
                            OutParts op, UnfCols out, const uint64_t* row_base,
                            uint8_t* values, uint8_t* ck_arena, uint8_t* static_vals,
                            uint8_t* cpx_bytes = nullptr, const uint64_t* cpx_base = nullptr) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t id = ids[sorted[i].idx];
    uint64_t ob = row_base[i];
    op.row_base[i] = ob;
    uint64_t cpx_cur = cpx_base ? cpx_base[i] : 0;
    if (gp.static_pct) {
        // oracle contract: gen_has_static / gen_static_ts; one blob cell
        bool has = splitmix64(gp.seed ^ 0x57A71CULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 <
                   gp.static_pct;
        if (has) {
            int64_t sts = gp.base_ts + (int64_t)(splitmix64(gp.seed ^ id * 977 ^ ((uint64_t)gp.sst << 48)) % 1000000000ULL);
            op.st.flags[i] = PF_HAS_ROW | PF_LIVE_TS;
            op.st.live_ts[i] = sts;
            op.st.live_ttl[i] = 0;
            op.st.live_let[i] = NO_DELETION_TIME;
            op.st.rdel_mfda[i] = INT64_MIN;
            op.st.rdel_ldt[i] = LDT_NONE_U32;
            op.st.cell_flags[i] = CELLF_PRESENT | CELLF_HAS_VALUE;
            op.st.cell_ts[i] = sts;
            op.st.cell_ldt[i] = LDT_NONE_U32;
            op.st.cell_ttl[i] = 0;
            op.st.val_addr[i] = (uint64_t)(static_vals + i * (uint64_t)gp.value_len);
            op.st.val_len[i] = gp.value_len;
        } else {
            op.st.flags[i] = 0;
            op.st.cell_flags[i] = 0;
        }
    }
    uint32_t emitted = 0;
    const uint32_t NCK = gp.ck_cols ? gp.ck_cols : (gp.clustering_rows ? 1 : 0);
    // ck encodes the full position; ck_cols==2 splits it as (ck/64, ck%64)
    // scaled by 16 per component (oracle gen_ck2). nv < NCK = prefix bound.
    auto put_ck = [&](uint64_t o, int64_t ck, bool has_ck, bool is_row, uint32_t nv) {
        if (!has_ck || NCK == 0) { out.ck_count[o] = 0; return; }
        out.ck_count[o] = (uint8_t)nv;
        for (uint32_t c = 0; c < nv; c++) {
            uint64_t oc = o * NCK + c;
            int64_t comp = NCK == 2 ? (c == 0 ? (ck / 64) * 16 : (ck % 64)) : ck;
            if (gp.ck_text) {
                uint8_t* dst = ck_arena + oc * 16;
                uint32_t len = gen2_ck_text(comp, is_row, dst);
                out.ck[oc] = ck_prefix_var(dst, len);
                out.ck_addr[oc] = (uint64_t)dst;
                out.ck_len[oc] = len;
            } else {
                out.ck[oc] = (uint64_t)comp ^ 0x8000000000000000ULL;
                out.ck_addr[oc] = 0;
                out.ck_len[oc] = 8;
            }
        }
    };
    const uint32_t NCV = gp.n_value_cols;
    auto put_marker = [&](uint8_t kind, int64_t ck, int64_t m, uint32_t l) {
        uint64_t o = ob + emitted++;
        for (uint32_t c = 0; c < NCV; c++) out.cell_flags[o * NCV + c] = 0;
        out.rkind[o] = kind;
        put_ck(o, ck, true, false, NCK == 2 ? 1 : NCK);  // 2-col mode: prefix bounds
        if (gp.complex_pct) {
            out.cpx_del_mfda[o] = INT64_MIN;
            out.cpx_del_ldt[o] = LDT_NONE_U32;
            out.cpx_start[o] = cpx_cur;
            out.cpx_count[o] = 0;
        }
        out.flags[o] = 0;
        out.live_ts[o] = NO_TIMESTAMP;
        out.live_ttl[o] = 0;
        out.live_let[o] = NO_DELETION_TIME;
        out.rdel_mfda[o] = m;
        out.rdel_ldt[o] = l;
        out.start_mfda[o] = INT64_MIN;
        out.start_ldt[o] = LDT_NONE_U32;
        out.cell_ts[o] = NO_TIMESTAMP;
        out.cell_ldt[o] = LDT_NONE_U32;
        out.cell_ttl[o] = 0;
        out.val_addr[o] = 0;
        out.val_len[o] = 0;
    };
    auto put_row = [&](int64_t ck, bool has_ck, int64_t ts, bool tomb, uint32_t tomb_ldt,
                       uint32_t rowj) {
        uint64_t o = ob + emitted++;
        out.rkind[o] = BK_CLUSTERING;
        put_ck(o, ck, has_ck, true, NCK);
        out.start_mfda[o] = INT64_MIN;
        out.start_ldt[o] = LDT_NONE_U32;
        out.live_ttl[o] = 0;
        out.live_let[o] = NO_DELETION_TIME;
        if (gp.complex_pct) {
            out.cpx_del_mfda[o] = INT64_MIN;
            out.cpx_del_ldt[o] = LDT_NONE_U32;
            out.cpx_start[o] = cpx_cur;
            out.cpx_count[o] = 0;
        }
        if (tomb) {
            out.flags[o] = PF_HAS_ROW | PF_ROW_DEL;
            out.live_ts[o] = NO_TIMESTAMP;
            out.rdel_mfda[o] = ts;
            out.rdel_ldt[o] = tomb_ldt;
            for (uint32_t c = 0; c < NCV; c++) out.cell_flags[o * NCV + c] = 0;
        } else {
            // oracle contract: gen_row_expiring / gen_ttl / gen_let —
            // expiring rows carry ExpiringLivenessInfo and their cells
            // ttl + localDeletionTime == localExpirationTime
            bool expg = gp.ttl_pct &&
                        splitmix64(gp.seed ^ 0x771E771EULL ^ id ^ ((uint64_t)gp.sst << 32) ^
                                   (uint64_t)(rowj + 3) * 101) % 100 < gp.ttl_pct;
            int32_t ettl = 0;
            int64_t elet = NO_DELETION_TIME;
            if (expg) {
                ettl = (int32_t)(60 + splitmix64(id ^ 0x77AA11ULL ^ (uint64_t)(rowj + 1) * 131) % 86400);
                elet = gp.base_ldt + (int64_t)(splitmix64(id ^ 0x1E7E1E7EULL ^ (uint64_t)(rowj + 1) * 17) % 2000);
                out.live_ttl[o] = ettl;
                out.live_let[o] = elet;
            }
            out.flags[o] = PF_HAS_ROW | PF_LIVE_TS;
            out.live_ts[o] = ts;
            out.rdel_mfda[o] = INT64_MIN;
            out.rdel_ldt[o] = LDT_NONE_U32;
            for (uint32_t c = 0; c < NCV; c++) {
                uint64_t oc = o * NCV + c;
                bool miss = gp.col_missing_pct &&
                            splitmix64(gp.seed ^ 0xC011C011ULL ^ id ^ ((uint64_t)(rowj + 1) << 40) ^
                                       ((uint64_t)(c + 1) << 56) ^ ((uint64_t)gp.sst << 32)) % 100 <
                                gp.col_missing_pct;
                if (miss) { out.cell_flags[oc] = 0; continue; }
                out.cell_flags[oc] = CELLF_PRESENT | CELLF_HAS_VALUE;
                out.cell_ts[oc] = ts;
                out.cell_ldt[oc] = expg ? (uint32_t)elet : LDT_NONE_U32;
                out.cell_ttl[oc] = expg ? ettl : 0;
                uint8_t* vp = values + oc * (uint64_t)gp.value_len;
                out.val_addr[oc] = (uint64_t)vp;
                if (gp.counter) {
                    // build the CounterContext in the value slot (oracle
                    // make_ctx mirror); stride gp.value_len >= 274
                    uint64_t cid = gp.clustering_rows ? (id ^ ((uint64_t)rowj << 20)) : id;
                    uint8_t body[8 * 32];
                    int16_t elts[8];
                    uint32_t nb = 0, ne = 0, shard_no = 0;
                    for (int pi2 = 0; pi2 < 8; pi2++) {
                        uint32_t idx = gp.ctr_pool_idx[pi2];
                        if (splitmix64(gp.seed ^ 0xC717C717ULL ^ cid ^ ((uint64_t)gp.sst << 32) ^
                                       (uint64_t)(idx + 1) * 37) % 100 >= 55) continue;
                        int role = idx % 3;
                        uint64_t r2 = splitmix64(gp.seed ^ 0xC10CULL ^ cid ^ (uint64_t)(idx + 1) * 131 ^
                                                 ((uint64_t)gp.sst << 40));
                        int64_t ck3;
                        if (role == 0) {
                            uint64_t base2 = splitmix64(gp.seed ^ 0x610BULL ^ cid ^ idx);
                            ck3 = (int64_t)(1000 + (cid % 2 ? base2 % 50 : (base2 + gp.sst) % 50));
                        } else if (role == 1) {
                            ck3 = (int64_t)(1 + r2 % 5);
                        } else if (idx == 7 && (cid % 5) == 0) {
                            ck3 = -(int64_t)(1 + r2 % 90);
                        } else {
                            ck3 = (int64_t)(1 + r2 % 99);
                        }
                        int64_t cn3 = (int64_t)(splitmix64(gp.seed ^ 0xC0117ULL ^ cid ^
                                                           (uint64_t)(idx + 1) * 17 ^
                                                           ((uint64_t)gp.sst << 24)) % 1000) - 100;
                        if (role == 0) elts[ne++] = (int16_t)((int32_t)shard_no + INT16_MIN);
                        else if (role == 1) elts[ne++] = (int16_t)shard_no;
                        for (int b = 0; b < 16; b++) body[nb + b] = gp.ctr_pool[pi2][b];
                        for (int b = 7; b >= 0; b--) body[nb + 16 + (7 - b)] = (uint8_t)((uint64_t)ck3 >> (8 * b));
                        for (int b = 7; b >= 0; b--) body[nb + 24 + (7 - b)] = (uint8_t)((uint64_t)cn3 >> (8 * b));
                        nb += 32;
                        shard_no++;
                    }
                    vp[0] = (uint8_t)(ne >> 8);
                    vp[1] = (uint8_t)ne;
                    for (uint32_t e2 = 0; e2 < ne; e2++) {
                        vp[2 + e2 * 2] = (uint8_t)((uint16_t)elts[e2] >> 8);
                        vp[3 + e2 * 2] = (uint8_t)elts[e2];
                    }
                    for (uint32_t b = 0; b < nb; b++) vp[2 + ne * 2 + b] = body[b];
                    out.val_len[oc] = 2 + ne * 2 + nb;
                } else {
                    out.val_len[oc] = gp.value_len;
                }
            }
            // complex column 'zm': dedup+sorted map cells (oracle put_complex)
            if (gp.complex_pct && gen2_has_cpx(gp, id, rowj)) {
                out.flags[o] |= PF_HAS_CPX;
                if (gen2_has_cpx_del(gp, id, rowj)) {
                    out.cpx_del_mfda[o] = ts - 1;
                    out.cpx_del_ldt[o] = gen2_ldt(gp, id * 5 + rowj, 0xCD);
                }
                uint32_t pv[4];
                uint32_t nc2 = gen2_cpx_paths(gp, id, rowj, pv);
                out.cpx_count[o] = nc2;
                for (uint32_t e = 0; e < nc2; e++) {
                    uint64_t xe = cpx_cur + e;
                    uint8_t* b = cpx_bytes + xe * 12;
                    uint32_t p4 = pv[e];
                    for (int bi = 0; bi < 4; bi++) b[bi] = (uint8_t)(p4 >> (8 * (3 - bi)));
                    uint64_t w = splitmix64(gp.seed ^ id * 131 ^ ((uint64_t)gp.sst << 40) ^
                                            (uint64_t)(rowj + 1) * 29 ^ (uint64_t)(p4 + 1) * 389);
                    for (int bi = 0; bi < 8; bi++) b[4 + bi] = (uint8_t)(w >> (8 * bi));
                    out.cpx.ts[xe] = ts - (int64_t)(p4 % 3);
                    out.cpx.ldt[xe] = LDT_NONE_U32;
                    out.cpx.ttl[xe] = 0;
                    out.cpx.flags[xe] = CELLF_PRESENT | CELLF_HAS_VALUE;
                    out.cpx.path_addr[xe] = (uint64_t)b;
                    out.cpx.path_len[xe] = 4;
                    out.cpx.val_addr[xe] = (uint64_t)(b + 4);
                    out.cpx.val_len[xe] = 8;
                }
                cpx_cur += nc2;
            }
        }
    };

    if (gp.clustering_rows == 0) {
        int64_t ts = gp.base_ts + (int64_t)(splitmix64(gp.seed ^ id * 31 ^ ((uint64_t)gp.sst << 48)) % 1000000000ULL);
        bool pdel = gp.partition_del_pct && (splitmix64(gp.seed ^ 0xFEEDULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 < gp.partition_del_pct);
        bool tomb = !pdel && gp.tombstone_pct && (splitmix64(gp.seed ^ 0xDEADULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 < gp.tombstone_pct);
        if (pdel) {
            op.pdel_mfda[i] = ts;
            op.pdel_ldt[i] = gen2_ldt(gp, id, 0xDD);
        } else {
            op.pdel_mfda[i] = INT64_MIN;
            op.pdel_ldt[i] = LDT_NONE_U32;
            put_row(0, false, ts, tomb, tomb ? gen2_ldt(gp, id, 0xEE) : 0, 0);
        }
    } else {
        op.pdel_mfda[i] = INT64_MIN;
        op.pdel_ldt[i] = LDT_NONE_U32;
        bool has_rt = gp.range_tomb_pct && (splitmix64(gp.seed ^ 0xBEEFULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 < gp.range_tomb_pct);
        int64_t rlo = 0, rhi = 0, rts = 0;
        uint32_t rldt = 0;
        if (has_rt) {
            uint64_t r = splitmix64(gp.seed ^ 0xB00BULL ^ id ^ ((uint64_t)gp.sst << 32));
            uint32_t a = (uint32_t)(r % gp.clustering_rows);
            uint32_t b = a + 1 + (uint32_t)((r >> 32) % (gp.clustering_rows - a));
            // 1-col: bounds between rows (a*16+8); 2-col: PREFIX bounds on the
            // first component covering whole ck0 groups (oracle gen mirror)
            rlo = gp.ck_cols == 2 ? (int64_t)(a / 4) * 64 : (int64_t)a * 16 + 8;
            rhi = gp.ck_cols == 2 ? (int64_t)(b / 4) * 64 + 63 : (int64_t)b * 16 + 8;
            rts = gp.base_ts + (int64_t)(splitmix64(gp.seed ^ 0xAB1EULL ^ id ^ ((uint64_t)gp.sst << 48)) % 1000000000ULL);
            rldt = gen2_ldt(gp, id, 0xCC);
        }
        bool rt_open = false, rt_done = false;
        for (uint32_t j = 0; j < gp.clustering_rows; j++) {
            int64_t ck = gp.ck_cols == 2 ? (int64_t)(j / 4) * 64 + (int64_t)(j % 4)
                                         : (int64_t)j * 16;
            // 2-col mode: the START bound is a ck0-group PREFIX and sorts
            // before the group's first row, so it opens on >= (1-col bounds
            // sit between rows and open on >)
            if (has_rt && !rt_open && !rt_done &&
                (gp.ck_cols == 2 ? ck >= rlo : ck > rlo)) {
                put_marker(BK_INCL_START, rlo, rts, rldt);
                rt_open = true;
            }
            int64_t ts = gp.base_ts + (int64_t)(splitmix64(gp.seed ^ id * 31 ^ ((uint64_t)gp.sst << 48) ^ (uint64_t)(j + 1) * 0x9E37ULL) % 1000000000ULL);
            bool tomb = gp.tombstone_pct && (splitmix64(gp.seed ^ 0xDEADULL ^ id ^ ((uint64_t)gp.sst << 32) ^ (uint64_t)(j + 7) * 131) % 100 < gp.tombstone_pct);
            put_row(ck, true, ts, tomb, tomb ? gen2_ldt(gp, id * 1000 + j, 0xEE) : 0, j);
            int64_t ck_next = gp.ck_cols == 2 ? (int64_t)((j + 1) / 4) * 64 + (int64_t)((j + 1) % 4)
                                              : (int64_t)(j + 1) * 16;
            if (has_rt && rt_open && j + 1 < gp.clustering_rows && ck_next > rhi) {
                put_marker(BK_INCL_END, rhi, rts, rldt);
                rt_open = false;
                rt_done = true;
            }
        }
        if (has_rt && rt_open) put_marker(BK_INCL_END, rhi, rts, rldt);
    }
    op.row_count[i] = emitted;
}

// values keyed like the oracle generator: simple schema seeds by key id; wide
// partitions seed by (id ^ (j<<52) ^ j)
__global__ void k_gen_values2(GenParams2 gp, const MRec* sorted, const uint64_t* ids,
                              OutParts op, UnfCols out, uint64_t n, uint8_t* values,
                              const uint64_t* row_base) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (gp.counter) return;  // counter contexts written by k_gen_fill2
    uint64_t id = ids[sorted[i].idx];
    if (gp.static_pct && (op.st.cell_flags[i] & CELLF_HAS_VALUE)) {
        uint64_t vid = id ^ 0xABCDEF57ULL;
        uint8_t* outp = (uint8_t*)op.st.val_addr[i];
        uint64_t state = gp.seed ^ vid * 0x100000001B3ULL ^ ((uint64_t)gp.sst << 40);
        uint64_t prev = splitmix64(state);
        uint32_t nw = (gp.value_len + 7) / 8;
        for (uint32_t w = 0; w < nw; w++) {
            uint64_t r = splitmix64(state + 1 + w);
            uint64_t word = (r % 100 < gp.value_repeat_pct && w > 0) ? prev : splitmix64(r);
            prev = word;
            uint32_t off = w * 8;
            for (uint32_t b = 0; b < 8 && off + b < gp.value_len; b++)
                outp[off + b] = (uint8_t)(word >> (8 * b));
        }
    }
    uint64_t rb = op.row_base[i];
    uint32_t cnt = op.row_count[i];
    uint32_t rowj = 0;
    for (uint32_t u = 0; u < cnt; u++) {
        uint64_t o = rb + u;
        if (out.rkind[o] != BK_CLUSTERING) continue;
        uint32_t j = rowj++;
        for (uint32_t c = 0; c < gp.n_value_cols; c++) {
        uint64_t oc = o * gp.n_value_cols + c;
        if (!(out.cell_flags[oc] & CELLF_HAS_VALUE)) continue;
        uint64_t seed_id = gp.clustering_rows == 0 ? id : (id ^ ((uint64_t)j << 52) ^ j);
        seed_id += (uint64_t)c * 0xA5A5A5A5A5A5A5ULL;
        uint8_t* outp = (uint8_t*)out.val_addr[oc];
        uint64_t state = gp.seed ^ seed_id * 0x100000001B3ULL ^ ((uint64_t)gp.sst << 40);
        uint64_t prev = splitmix64(state);
        uint32_t nw = (gp.value_len + 7) / 8;
        for (uint32_t w = 0; w < nw; w++) {
            uint64_t r = splitmix64(state + 1 + w);
            uint64_t word = (r % 100 < gp.value_repeat_pct && w > 0) ? prev : splitmix64(r);
            prev = word;
            uint32_t off = w * 8;
            for (uint32_t b = 0; b < 8 && off + b < gp.value_len; b++)
                outp[off + b] = (uint8_t)(word >> (8 * b));
        }
        }
    }
    (void)values;
    (void)row_base;
}

}  // namespace gpuc
