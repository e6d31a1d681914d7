// PRODUCT — CDNA4 (gfx950) kernels for the SSTable compaction pipeline.
// All byte-level compute of the CompactionTask hot loop runs here:
// chunk LZ4 decode + CRC, row decode, k-way merge, reconcile+purge,
// re-serialize, chunk LZ4 encode + CRC, bloom. Wavefront size 64.
#include <hip/hip_runtime.h>
#include "codec.h"
#include "gpu_structs.h"
#include "lz4_model.h"

namespace gpuc {

#define WAVE 64

// ---------------------------------------------------------------------------
// CRC32 slice-by-8 tables (host-filled once per process, device-resident)
// ---------------------------------------------------------------------------
__device__ uint32_t g_crc_tab8[8][256];

__global__ void k_crc_init() {
    // single block builds tables
    int t = threadIdx.x;
    for (int i = t; i < 256; i += blockDim.x) {
        uint32_t c = (uint32_t)i;
        for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
        g_crc_tab8[0][i] = c;
    }
    __syncthreads();
    for (int s = 1; s < 8; s++) {
        for (int i = t; i < 256; i += blockDim.x) {
            uint32_t c = g_crc_tab8[s - 1][i];
            g_crc_tab8[s][i] = g_crc_tab8[0][c & 0xFF] ^ (c >> 8);
        }
        __syncthreads();
    }
}

__device__ inline uint32_t d_crc32(const uint8_t* p, uint32_t len) {
    uint32_t crc = 0xFFFFFFFFu;
    // byte loop with slice-by-1 (device table); adequate next to LZ4 cost
    for (uint32_t i = 0; i < len; i++) crc = g_crc_tab8[0][(crc ^ p[i]) & 0xFF] ^ (crc >> 8);
    return ~crc;
}

// ---------------------------------------------------------------------------
// LZ4 chunk decode (one wave per chunk; lane 0 drives, lanes assist copies)
// frame: 4-byte LE uncompressed length + raw LZ4 block, then 4-byte BE CRC of
// the compressed bytes (CompressedSequentialWriter.flushData + LZ4Compressor)
// ---------------------------------------------------------------------------
struct ChunkDesc {
    const uint8_t* comp;  // at the 4-byte LE length header
    uint8_t* out;
    uint32_t comp_len;    // excluding trailing CRC
    uint32_t out_len;
};

__global__ void k_lz4_decompress(const ChunkDesc* chunks, uint32_t n, int verify_crc,
                                 unsigned long long* error) {
    uint32_t c = blockIdx.x * (blockDim.x / WAVE) + (threadIdx.x / WAVE);
    if (c >= n) return;
    int lane = threadIdx.x & (WAVE - 1);
    ChunkDesc ch = chunks[c];
    if (verify_crc && lane == 0) {
        uint32_t stored = ((uint32_t)ch.comp[ch.comp_len] << 24) | ((uint32_t)ch.comp[ch.comp_len + 1] << 16) |
                          ((uint32_t)ch.comp[ch.comp_len + 2] << 8) | ch.comp[ch.comp_len + 3];
        if (d_crc32(ch.comp, ch.comp_len) != stored) atomicExch(error, 1ull);
    }
    uint32_t hdr = ch.comp[0] | (ch.comp[1] << 8) | (ch.comp[2] << 16) | ((uint32_t)ch.comp[3] << 24);
    if (hdr != ch.out_len) { if (lane == 0) atomicExch(error, 2ull); return; }
    // lane0-sequential token parse; all lanes cooperate on copies via shfl of state
    if (lane != 0) return;  // round-1: lane0 decode (wave-coop decode is a later optimization)
    const uint8_t* ip = ch.comp + 4;
    const uint8_t* iend = ch.comp + ch.comp_len;
    uint8_t* op = ch.out;
    uint8_t* oend = ch.out + ch.out_len;
    while (ip < iend) {
        uint32_t token = *ip++;
        uint32_t lit = token >> 4;
        if (lit == 15) { uint32_t s; do { s = *ip++; lit += s; } while (s == 255); }
        if (op + lit > oend || ip + lit > iend) { atomicExch(error, 3ull); return; }
        for (uint32_t i = 0; i < lit; i++) op[i] = ip[i];
        op += lit; ip += lit;
        if (ip >= iend) break;  // last literals
        uint32_t off = ip[0] | (ip[1] << 8);
        ip += 2;
        uint32_t ml = (token & 15) + 4;
        if (ml == 19) { uint32_t s; do { s = *ip++; ml += s; } while (s == 255); }
        if (off == 0 || op - ch.out < (ptrdiff_t)off || op + ml > oend) { atomicExch(error, 4ull); return; }
        const uint8_t* m = op - off;
        for (uint32_t i = 0; i < ml; i++) op[i] = m[i];  // overlap-correct byte copy
        op += ml;
    }
    if (op != oend) atomicExch(error, 5ull);
}

// ---------------------------------------------------------------------------
// partition parse (one thread per input partition)
// Row wire format: UnfilteredSerializer.java:36-117 (+ Cell.java:240-306);
// partition frame: SortedTablePartitionWriter.java:97-166.
// ---------------------------------------------------------------------------
enum : uint8_t {
    F_END = 0x01, F_MARKER = 0x02, F_TS = 0x04, F_TTL = 0x08, F_DEL = 0x10,
    F_ALLCOL = 0x20, F_COMPLEX = 0x40, F_EXT = 0x80
};
enum : uint8_t { CF_DELETED = 1, CF_EXPIRING = 2, CF_EMPTY = 4, CF_ROWTS = 8, CF_ROWTTL = 16 };

struct SrcDesc {
    const uint8_t* data;       // decompressed image
    const uint64_t* part_pos;  // n_parts+1 offsets (last = data_len)
    uint32_t n_parts;
    int64_t min_ts, min_ldt;   // this source's EncodingStats bases
    int32_t min_ttl;
    uint32_t rec_base;         // offset into the concatenated arrays
};

__global__ void k_parse(const SrcDesc* srcs, uint32_t n_srcs, uint32_t total,
                        MRec* recs, ParsedCols pc, int32_t col_fixed_len,
                        unsigned long long* error, unsigned long long* rows_in) {
    uint32_t gi = blockIdx.x * blockDim.x + threadIdx.x;
    if (gi >= total) return;
    // locate source
    uint32_t s = 0;
    while (s + 1 < n_srcs && gi >= srcs[s + 1].rec_base) s++;
    const SrcDesc& sd = srcs[s];
    uint32_t li = gi - sd.rec_base;
    const uint8_t* base = sd.data;
    uint64_t pos = sd.part_pos[li];
    uint64_t end = sd.part_pos[li + 1];

    uint32_t klen = ((uint32_t)base[pos] << 8) | base[pos + 1];
    pos += 2;
    if (klen == 0 || klen > 8) { atomicExch(error, 10ull); return; }
    uint64_t pfx = 0;
    for (uint32_t b = 0; b < klen; b++) pfx |= (uint64_t)base[pos + b] << (8 * (7 - b));
    uint8_t keybytes[8];
    for (uint32_t b = 0; b < klen; b++) keybytes[b] = base[pos + b];
    pos += klen;
    int64_t token = murmur3_token(keybytes, klen);

    MRec r;
    r.tok = (uint64_t)token ^ 0x8000000000000000ULL;
    r.pfx = pfx;
    r.idx = li;
    r.src = (uint16_t)s;
    r.klen = (uint8_t)klen;
    r.pad = 0;
    recs[gi] = r;

    // partition deletion (DeletionTime oa serializer)
    int64_t pdel_mfda = INT64_MIN;
    uint32_t pdel_ldt = LDT_NONE_U32;
    {
        uint8_t f = base[pos];
        if (f & 0x80) { pos++; if (f != 0x80) { atomicExch(error, 11ull); return; } }
        else {
            uint64_t v = 0;
            for (int i = 0; i < 8; i++) v = (v << 8) | base[pos + i];
            pdel_mfda = (int64_t)v;
            pdel_ldt = ((uint32_t)base[pos + 8] << 24) | ((uint32_t)base[pos + 9] << 16) |
                       ((uint32_t)base[pos + 10] << 8) | base[pos + 11];
            pos += 12;
        }
    }
    pc.pdel_mfda[gi] = pdel_mfda;
    pc.pdel_ldt[gi] = pdel_ldt;

    uint8_t pf = 0;
    int64_t lts = NO_TIMESTAMP, llet = NO_DELETION_TIME, rmfda = INT64_MIN;
    int32_t lttl = 0;
    uint32_t rldt = LDT_NONE_U32;
    int64_t cts = NO_TIMESTAMP;
    uint32_t cldt = LDT_NONE_U32;
    int32_t cttl = 0;
    uint64_t vaddr = 0;
    uint32_t vlen = 0;

    uint8_t flags = base[pos++];
    if (!(flags & F_END)) {
        if (flags & (F_MARKER | F_EXT | F_COMPLEX)) { atomicExch(error, 12ull); return; }
        pf |= PF_HAS_ROW;
        // no clustering columns; row size + prev size vints (skip-aids)
        uvint_get(base, &pos);
        uvint_get(base, &pos);
        if (flags & F_TS) { pf |= PF_LIVE_TS; lts = (int64_t)uvint_get(base, &pos) + sd.min_ts; }
        if (flags & F_TTL) {
            lttl = (int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ttl;
            llet = (int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt;
        }
        if (flags & F_DEL) {
            pf |= PF_ROW_DEL;
            rmfda = (int64_t)uvint_get(base, &pos) + sd.min_ts;
            rldt = ldt_u32((int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt);
        }
        bool has_cell;
        if (flags & F_ALLCOL) has_cell = true;
        else {
            uint64_t missing = uvint_get(base, &pos);  // bitmap of missing columns (1 column)
            has_cell = !(missing & 1);
        }
        if (has_cell) {
            pf |= PF_HAS_CELL;
            uint8_t cf = base[pos++];
            cts = (cf & CF_ROWTS) ? lts : (int64_t)uvint_get(base, &pos) + sd.min_ts;
            bool dead = cf & CF_DELETED, exp = cf & CF_EXPIRING;
            int64_t ldtl;
            if (cf & CF_ROWTTL) ldtl = llet;
            else if (dead || exp) ldtl = (int64_t)(int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ldt;
            else ldtl = NO_DELETION_TIME;
            cttl = (cf & CF_ROWTTL) ? lttl : (exp ? (int32_t)(uint32_t)uvint_get(base, &pos) + sd.min_ttl : 0);
            cldt = ldt_u32(ldtl);
            if (exp) pf |= PF_CELL_EXPIRING;
            if (!(cf & CF_EMPTY)) {
                pf |= PF_CELL_VALUE;
                vlen = col_fixed_len >= 0 ? (uint32_t)col_fixed_len : (uint32_t)uvint_get(base, &pos);
                vaddr = (uint64_t)(base + pos);
                pos += vlen;
            }
        }
        uint8_t nxt = base[pos++];
        if (!(nxt & F_END)) { atomicExch(error, 13ull); return; }  // >1 row unsupported (no clustering)
        atomicAdd(rows_in, 1ull);
    }
    if (pos != end) { atomicExch(error, 14ull); return; }
    pc.flags[gi] = pf;
    pc.live_ts[gi] = lts;
    pc.live_ttl[gi] = lttl;
    pc.live_let[gi] = llet;
    pc.rdel_mfda[gi] = rmfda;
    pc.rdel_ldt[gi] = rldt;
    pc.cell_ts[gi] = cts;
    pc.cell_ldt[gi] = cldt;
    pc.cell_ttl[gi] = cttl;
    pc.val_addr[gi] = vaddr;
    pc.val_len[gi] = vlen;
}

// ---------------------------------------------------------------------------
// pairwise stable merge (merge path, per-thread tiles)
// ---------------------------------------------------------------------------
struct MergePair {
    uint64_t a_beg, a_end, b_beg, b_end, out_beg;
    uint64_t tile_beg;  // first global tile of this pair
};

#define MERGE_TILE 16

__global__ void k_merge_pairs(const MRec* in, MRec* out, const MergePair* pairs,
                              uint32_t n_pairs, uint64_t total_tiles) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= total_tiles) return;
    // find pair by binary search over tile_beg
    uint32_t lo = 0, hi = n_pairs - 1;
    while (lo < hi) {
        uint32_t mid = (lo + hi + 1) >> 1;
        if (pairs[mid].tile_beg <= t) lo = mid; else hi = mid - 1;
    }
    const MergePair p = pairs[lo];
    uint64_t an = p.a_end - p.a_beg, bn = p.b_end - p.b_beg;
    uint64_t k = (t - p.tile_beg) * MERGE_TILE;  // output offset within pair
    uint64_t kend = k + MERGE_TILE;
    uint64_t n_out = an + bn;
    if (k >= n_out) return;
    if (kend > n_out) kend = n_out;
    // diagonal search: find (i,j), i+j=k, partitioning with stability (A first on ties):
    // largest i such that A[i-1] <= B[k-i]  (i.e. !less(B[k-i], A[i-1]))
    uint64_t ilo = k > bn ? k - bn : 0;
    uint64_t ihi = k < an ? k : an;
    while (ilo < ihi) {
        uint64_t i = (ilo + ihi + 1) >> 1;
        // A[i-1] goes before B[k-i] iff !(B < A) — ties favor A
        if (!mrec_less(in[p.b_beg + (k - i)], in[p.a_beg + i - 1])) ilo = i;
        else ihi = i - 1;
    }
    uint64_t i = ilo, j = k - ilo;
    for (uint64_t o = k; o < kend; o++) {
        bool take_a;
        if (i >= an) take_a = false;
        else if (j >= bn) take_a = true;
        else take_a = !mrec_less(in[p.b_beg + j], in[p.a_beg + i]);  // ties -> A
        out[p.out_beg + o] = take_a ? in[p.a_beg + i++] : in[p.b_beg + j++];
    }
}

// ---------------------------------------------------------------------------
// group heads + exclusive scan utilities (u64)
// ---------------------------------------------------------------------------
__global__ void k_group_heads(const MRec* recs, uint64_t n, uint64_t* head) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    head[i] = (i == 0) || !mrec_eq(recs[i], recs[i - 1]);
}

#define SCAN_BLOCK 256
#define SCAN_ITEMS 8
#define SCAN_TILE (SCAN_BLOCK * SCAN_ITEMS)

__global__ void k_scan_partial(const uint64_t* in, uint64_t* out, uint64_t* block_sums, uint64_t n) {
    __shared__ uint64_t sh[SCAN_BLOCK];
    uint64_t base = (uint64_t)blockIdx.x * SCAN_TILE;
    uint64_t v[SCAN_ITEMS];
    uint64_t sum = 0;
    for (int k = 0; k < SCAN_ITEMS; k++) {
        uint64_t idx = base + threadIdx.x * SCAN_ITEMS + k;
        v[k] = idx < n ? in[idx] : 0;
        sum += v[k];
    }
    sh[threadIdx.x] = sum;
    __syncthreads();
    // Hillis-Steele inclusive scan over the block
    for (int off = 1; off < SCAN_BLOCK; off <<= 1) {
        uint64_t x = threadIdx.x >= (uint32_t)off ? sh[threadIdx.x - off] : 0;
        __syncthreads();
        sh[threadIdx.x] += x;
        __syncthreads();
    }
    uint64_t excl = sh[threadIdx.x] - sum;
    for (int k = 0; k < SCAN_ITEMS; k++) {
        uint64_t idx = base + threadIdx.x * SCAN_ITEMS + k;
        if (idx < n) out[idx] = excl;
        excl += v[k];
    }
    if (threadIdx.x == SCAN_BLOCK - 1 && block_sums) block_sums[blockIdx.x] = sh[SCAN_BLOCK - 1];
}

__global__ void k_scan_add(uint64_t* data, const uint64_t* block_offsets, uint64_t n) {
    uint64_t base = (uint64_t)blockIdx.x * SCAN_TILE;
    uint64_t off = block_offsets[blockIdx.x];
    for (int k = 0; k < SCAN_ITEMS; k++) {
        uint64_t idx = base + threadIdx.x + (uint64_t)k * SCAN_BLOCK;
        if (idx < n) data[idx] += off;
    }
}

// group starts: scatter positions where head==1; group_of[i] for members
__global__ void k_group_starts(const uint64_t* head, const uint64_t* head_scan, uint64_t n,
                               uint64_t* group_start, uint64_t* n_groups) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (head[i]) group_start[head_scan[i]] = i;
    if (i == n - 1) *n_groups = head_scan[i] + 1;
}

// ---------------------------------------------------------------------------
// reconcile + purge (one thread per group)
// Row.Merger (Row.java:730-781) + Cells.reconcile (Cells.java:68-119) +
// PurgeFunction (PurgeFunction.java:26-145) + purge evaluator
// (CompactionController.java:247-286), simple schema (≤1 row, 1 column).
// ---------------------------------------------------------------------------
struct PurgeParams {
    int64_t now_sec, gc_before;
    int32_t never_purge;
    int32_t enforce_strict_liveness;
    const int64_t* ov_lo;      // overlap ranges (token intervals), sorted by lo
    const int64_t* ov_hi;
    const int64_t* ov_min_ts;
    int32_t n_overlaps;
    int32_t has_shard;
    int64_t shard_lo, shard_hi;
};

__device__ inline bool purge_evaluator(const PurgeParams& pp, int64_t token, int64_t ts) {
    int64_t min_ts = INT64_MAX;
    bool has = false;
    for (int i = 0; i < pp.n_overlaps; i++) {
        if (token >= pp.ov_lo[i] && token <= pp.ov_hi[i]) { has = true; min_ts = min(min_ts, pp.ov_min_ts[i]); }
    }
    return !has || ts < min_ts;
}
__device__ inline bool should_purge(const PurgeParams& pp, int64_t token, int64_t ts, int64_t ldt) {
    if (pp.never_purge) return false;
    return ldt < pp.gc_before && purge_evaluator(pp, token, ts);
}
// DeletionTime.supersedes (DeletionTime.java:158-161)
__device__ inline bool dt_supersedes(int64_t am, uint32_t al, int64_t bm, uint32_t bl) {
    return am > bm || (am == bm && ldt_long(al) > ldt_long(bl));
}
// compareValues(left,right) < 0 — unsigned lexicographic (ValueAccessor.compare)
__device__ inline bool value_less(uint64_t la, uint32_t ll, uint64_t ra, uint32_t rl) {
    const uint8_t* l = (const uint8_t*)la;
    const uint8_t* r = (const uint8_t*)ra;
    uint32_t n = ll < rl ? ll : rl;
    for (uint32_t i = 0; i < n; i++) {
        if (l[i] != r[i]) return l[i] < r[i];
    }
    return ll < rl;
}
// output-stats atomics (i64 via sign-flip to unsigned)
__device__ inline void stat_ts(OutStats* st, int64_t ts) {
    if (ts == NO_TIMESTAMP) return;
    unsigned long long v = (unsigned long long)ts ^ 0x8000000000000000ULL;
    atomicMin(&st->min_ts_flip, v);
    atomicMax(&st->max_ts_flip, v);
}
__device__ inline void stat_ldt(OutStats* st, int64_t l) {
    unsigned long long v = (unsigned long long)l ^ 0x8000000000000000ULL;
    atomicMin(&st->min_ldt_flip, v);
    atomicMax(&st->max_ldt_flip, v);
}
__device__ inline void stat_ttl(OutStats* st, int32_t ttl) {
    atomicMin(&st->min_ttl, (unsigned int)ttl);
    atomicMax(&st->max_ttl, (unsigned int)ttl);
}
__device__ inline void tomb_push(OutStats* st, uint32_t* ldts, uint32_t cap, uint32_t ldt) {
    unsigned long long i = atomicAdd(&st->tomb_count, 1ull);
    if (i < cap) ldts[i] = ldt;
}

__global__ void k_reconcile(const MRec* recs, const uint64_t* group_start, uint64_t n_groups,
                            uint64_t n_recs, const uint32_t* src_bases, ParsedCols pc,
                            OutParts op, PurgeParams pp, OutStats* st,
                            uint32_t* tomb_ldts, uint32_t tomb_cap) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    uint64_t beg = group_start[g];
    uint64_t endi = g + 1 < n_groups ? group_start[g + 1] : n_recs;
    uint32_t k = (uint32_t)(endi - beg);
    atomicAdd(&st->merged_counts[k > 64 ? 63 : k - 1], 1ull);

    const MRec r0 = recs[beg];
    int64_t token = (int64_t)(r0.tok ^ 0x8000000000000000ULL);
    op.keypfx[g] = r0.pfx;
    op.klen[g] = r0.klen;

    if (pp.has_shard && (token < pp.shard_lo || token > pp.shard_hi)) { op.keep[g] = 0; return; }

    // ---- merge partition deletion (UnfilteredRowIterators.java:465-482) ----
    int64_t pdm = INT64_MIN;
    uint32_t pdl = LDT_NONE_U32;
    for (uint64_t m = beg; m < endi; m++) {
        uint32_t i = src_bases[recs[m].src] + recs[m].idx;
        if (!dt_supersedes(pdm, pdl, pc.pdel_mfda[i], pc.pdel_ldt[i])) {
            pdm = pc.pdel_mfda[i];
            pdl = pc.pdel_ldt[i];
        }
    }

    // ---- merge the (single-clustering) row versions ----
    // Row.Merger.merge(activeDeletion = merged partition deletion)
    uint8_t of = 0;
    int64_t lts = NO_TIMESTAMP, llet = NO_DELETION_TIME;
    int32_t lttl = 0;
    int64_t rdm = INT64_MIN;
    uint32_t rdl = LDT_NONE_U32;
    int64_t cts = NO_TIMESTAMP;
    uint32_t cldt = LDT_NONE_U32;
    int32_t cttl = 0;
    uint64_t vaddr = 0;
    uint32_t vlen = 0;
    bool have_cell = false, cell_has_value = false, cell_expiring = false;

    uint32_t row_versions = 0;
    uint32_t last_row_i = 0;
    for (uint64_t m = beg; m < endi; m++) {
        uint32_t i = src_bases[recs[m].src] + recs[m].idx;
        if (pc.flags[i] & PF_HAS_ROW) { row_versions++; last_row_i = i; }
    }
    if (row_versions > 0) {
        bool pdel_live0 = (pdm == INT64_MIN && pdl == LDT_NONE_U32);
        // k==1: UnfilteredRowIterators.merge of a single iterator returns it
        // unchanged (no shadow filtering against its own partition deletion);
        // row_versions==1 under a LIVE active deletion is the Row.Merger
        // single-row shortcut (Row.java:733-741). Both are passthrough.
        if (k == 1 || (row_versions == 1 && pdel_live0)) {
            uint32_t i = last_row_i;
            uint8_t f = pc.flags[i];
            of = f & (PF_HAS_ROW | PF_LIVE_TS | PF_ROW_DEL | PF_HAS_CELL | PF_CELL_VALUE | PF_CELL_EXPIRING);
            lts = pc.live_ts[i]; lttl = pc.live_ttl[i]; llet = pc.live_let[i];
            rdm = pc.rdel_mfda[i]; rdl = pc.rdel_ldt[i];
            cts = pc.cell_ts[i]; cldt = pc.cell_ldt[i]; cttl = pc.cell_ttl[i];
            vaddr = pc.val_addr[i]; vlen = pc.val_len[i];
            have_cell = f & PF_HAS_CELL;
            cell_has_value = f & PF_CELL_VALUE;
            cell_expiring = f & PF_CELL_EXPIRING;
        } else {
            // liveness supersedes (LivenessInfo.supersedes) and deletion supersedes
            bool has_live = false;
            for (uint64_t m = beg; m < endi; m++) {
                uint32_t i = src_bases[recs[m].src] + recs[m].idx;
                uint8_t f = pc.flags[i];
                if (!(f & PF_HAS_ROW)) continue;
                if (f & PF_LIVE_TS) {
                    int64_t t2 = pc.live_ts[i];
                    int32_t ttl2 = pc.live_ttl[i];
                    int64_t let2 = pc.live_let[i];
                    bool sup;
                    if (!has_live) sup = true;
                    else if (t2 != lts) sup = t2 > lts;
                    else {
                        bool e1 = lttl == INT32_MAX, e2 = ttl2 == INT32_MAX;  // EXPIRED_LIVENESS_TTL
                        if (e1 != e2) sup = e2;
                        else if ((lttl != 0) == (ttl2 != 0)) sup = let2 > llet;
                        else sup = ttl2 != 0;
                    }
                    if (sup) { lts = t2; lttl = ttl2; llet = let2; has_live = true; }
                }
                if (f & PF_ROW_DEL) {
                    if (dt_supersedes(pc.rdel_mfda[i], pc.rdel_ldt[i], rdm, rdl)) {
                        rdm = pc.rdel_mfda[i];
                        rdl = pc.rdel_ldt[i];
                    }
                }
            }
            if (has_live) of |= PF_LIVE_TS;
            // active deletion: partition vs row deletion (Row.java:757-762)
            int64_t am = pdm;
            uint32_t al = pdl;
            bool row_del_kept = false;
            if (dt_supersedes(rdm, rdl, am, al)) { am = rdm; al = rdl; row_del_kept = true; }
            if (!row_del_kept) { rdm = INT64_MIN; rdl = LDT_NONE_U32; }
            else of |= PF_ROW_DEL;
            // activeDeletion.deletes(liveness)
            if (has_live && lts <= am) { of &= ~PF_LIVE_TS; lts = NO_TIMESTAMP; lttl = 0; llet = NO_DELETION_TIME; }
            // cell reconcile in source order (ColumnDataReducer + Cells.reconcile)
            for (uint64_t m = beg; m < endi; m++) {
                uint32_t i = src_bases[recs[m].src] + recs[m].idx;
                uint8_t f = pc.flags[i];
                if (!(f & PF_HAS_ROW) || !(f & PF_HAS_CELL)) continue;
                int64_t ts2 = pc.cell_ts[i];
                if (ts2 <= am) continue;  // activeDeletion.deletes(cell)
                if (!have_cell) {
                    have_cell = true;
                    cts = ts2; cldt = pc.cell_ldt[i]; cttl = pc.cell_ttl[i];
                    vaddr = pc.val_addr[i]; vlen = pc.val_len[i];
                    cell_has_value = f & PF_CELL_VALUE;
                    cell_expiring = f & PF_CELL_EXPIRING;
                    continue;
                }
                // reconcile(existing=left, candidate=right) — Cells.java:79-119
                bool take_right = false;
                uint32_t rl = pc.cell_ldt[i];
                bool l_dt = cldt != LDT_NONE_U32, r_dt = rl != LDT_NONE_U32;
                if (cts != ts2) take_right = ts2 > cts;
                else if (l_dt || r_dt) {
                    if (l_dt != r_dt) take_right = r_dt;
                    else {
                        bool l_tomb = !cell_expiring, r_tomb = !(pc.flags[i] & PF_CELL_EXPIRING);
                        if (l_tomb != r_tomb) take_right = r_tomb;
                        else if (cldt != rl) take_right = ldt_long(rl) > ldt_long(cldt);
                        else take_right = value_less(vaddr, vlen, pc.val_addr[i], pc.val_len[i]);
                    }
                } else {
                    take_right = value_less(vaddr, vlen, pc.val_addr[i], pc.val_len[i]);
                }
                if (take_right) {
                    cts = ts2; cldt = rl; cttl = pc.cell_ttl[i];
                    vaddr = pc.val_addr[i]; vlen = pc.val_len[i];
                    cell_has_value = pc.flags[i] & PF_CELL_VALUE;
                    cell_expiring = pc.flags[i] & PF_CELL_EXPIRING;
                }
            }
            if (have_cell) of |= PF_HAS_CELL;
            if (cell_has_value) of |= PF_CELL_VALUE;
            if (cell_expiring) of |= PF_CELL_EXPIRING;
            if ((of & (PF_LIVE_TS | PF_ROW_DEL | PF_HAS_CELL)) != 0) of |= PF_HAS_ROW;
        }
    }

    // ---- purge (PurgeFunction + BTreeRow.purge + AbstractCell.purge) ----
    bool pdel_live = (pdm == INT64_MIN && pdl == LDT_NONE_U32);
    if (!pdel_live && should_purge(pp, token, pdm, ldt_long(pdl))) { pdm = INT64_MIN; pdl = LDT_NONE_U32; pdel_live = true; }
    if (of & PF_HAS_ROW) {
        if (of & PF_LIVE_TS) {
            // DeletionPurger.shouldPurge(liveness, now): !isLive(now) && purge(ts, let)
            bool is_live = lttl == INT32_MAX ? false : (lttl != 0 ? pp.now_sec < llet : true);
            if (!is_live && should_purge(pp, token, lts, llet)) { of &= ~PF_LIVE_TS; lts = NO_TIMESTAMP; lttl = 0; llet = NO_DELETION_TIME; }
        }
        if ((of & PF_ROW_DEL) && should_purge(pp, token, rdm, ldt_long(rdl))) { of &= ~PF_ROW_DEL; rdm = INT64_MIN; rdl = LDT_NONE_U32; }
        if (of & PF_HAS_CELL) {
            // AbstractCell.purge (AbstractCell.java:78-99)
            bool live_cell = cldt == LDT_NONE_U32 || (cttl != 0 && pp.now_sec < ldt_long(cldt));
            if (!live_cell) {
                if (should_purge(pp, token, cts, ldt_long(cldt))) { of &= ~(PF_HAS_CELL | PF_CELL_VALUE | PF_CELL_EXPIRING); }
                else if (cttl != 0) {
                    // expired cell -> tombstone (value dropped, ldt -= ttl), purge again
                    int64_t nldt = ldt_long(cldt) - cttl;
                    if (should_purge(pp, token, cts, nldt)) { of &= ~(PF_HAS_CELL | PF_CELL_VALUE | PF_CELL_EXPIRING); }
                    else {
                        cldt = ldt_u32(nldt);
                        cttl = 0;
                        of &= ~(PF_CELL_VALUE | PF_CELL_EXPIRING);
                        vlen = 0;
                    }
                }
            }
        }
        if (!(of & (PF_LIVE_TS | PF_ROW_DEL | PF_HAS_CELL))) of = 0;  // row purged to nothing
        else if (pp.enforce_strict_liveness && !(of & PF_LIVE_TS) && !(of & PF_ROW_DEL)) of = 0;
    }
    bool keep = !(pdel_live && !(of & PF_HAS_ROW));
    op.pdel_mfda[g] = pdm;
    op.pdel_ldt[g] = pdl;
    op.flags[g] = of;
    op.live_ts[g] = lts;
    op.live_ttl[g] = lttl;
    op.live_let[g] = llet;
    op.rdel_mfda[g] = rdm;
    op.rdel_ldt[g] = rdl;
    op.cell_ts[g] = cts;
    op.cell_ldt[g] = cldt;
    op.cell_ttl[g] = cttl;
    op.val_addr[g] = vaddr;
    op.val_len[g] = (of & PF_CELL_VALUE) ? vlen : 0;
    op.keep[g] = keep ? 1 : 0;

}

// ---------------------------------------------------------------------------
// output stats collection over the final OutParts (MetadataCollector
// semantics, spec'd identically in oracle/src/sstable.cpp write_sstable)
// ---------------------------------------------------------------------------
__global__ void k_collect_outstats(OutParts op, uint64_t n, OutStats* st,
                                   uint32_t* tomb_ldts, uint32_t tomb_cap) {
    __shared__ unsigned long long sh_parts, sh_rows, sh_cells, sh_mints, sh_maxts,
        sh_minldt, sh_maxldt, sh_first, sh_last, sh_haspdel;
    __shared__ unsigned int sh_minttl, sh_maxttl;
    if (threadIdx.x == 0) {
        sh_parts = sh_rows = sh_cells = 0;
        sh_mints = sh_minldt = 0xFFFFFFFFFFFFFFFFULL;
        sh_maxts = sh_maxldt = 0;
        sh_first = 0xFFFFFFFFFFFFFFFFULL;
        sh_last = 0;
        sh_haspdel = 0;
        sh_minttl = 0xFFFFFFFFu;
        sh_maxttl = 0;
    }
    __syncthreads();
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g < n && op.keep[g]) {
        uint8_t of = op.flags[g];
        atomicAdd(&sh_parts, 1ull);
        if (of & PF_HAS_ROW) atomicAdd(&sh_rows, 1ull);
        atomicMin(&sh_first, (unsigned long long)g);
        atomicMax(&sh_last, (unsigned long long)g);
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
        bool pdel_live = op.pdel_mfda[g] == INT64_MIN && op.pdel_ldt[g] == LDT_NONE_U32;
        if (!pdel_live) {
            atomicExch(&sh_haspdel, 1ull);
            lts(op.pdel_mfda[g]);
            lldt(ldt_long(op.pdel_ldt[g]));
            tomb_push(st, tomb_ldts, tomb_cap, op.pdel_ldt[g]);
        }
        if (of & PF_LIVE_TS) {
            lts(op.live_ts[g]);
            if (op.live_ttl[g] != 0) {
                lldt(op.live_let[g]);
                atomicMin(&sh_minttl, (unsigned)op.live_ttl[g]);
                atomicMax(&sh_maxttl, (unsigned)op.live_ttl[g]);
            } else lldt(NO_DELETION_TIME);
        }
        if (of & PF_ROW_DEL) {
            lts(op.rdel_mfda[g]);
            lldt(ldt_long(op.rdel_ldt[g]));
            tomb_push(st, tomb_ldts, tomb_cap, op.rdel_ldt[g]);
        }
        if (of & PF_HAS_CELL) {
            atomicAdd(&sh_cells, 1ull);
            lts(op.cell_ts[g]);
            uint32_t cldt = op.cell_ldt[g];
            int32_t cttl = op.cell_ttl[g];
            if (cldt != LDT_NONE_U32 && cttl == 0) { lldt(ldt_long(cldt)); tomb_push(st, tomb_ldts, tomb_cap, cldt); }
            else if (cttl != 0) {
                lldt(ldt_long(cldt));
                atomicMin(&sh_minttl, (unsigned)cttl);
                atomicMax(&sh_maxttl, (unsigned)cttl);
            } else lldt(NO_DELETION_TIME);
        }
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
        atomicMin(&st->first_group, sh_first);
        atomicMax(&st->last_group, sh_last);
        if (sh_haspdel) atomicExch(&st->has_partition_deletions, 1ull);
    }
}

// group starts using the scanned head values (heads recomputed from recs)
__global__ void k_group_starts2(const MRec* recs, uint64_t n, const uint64_t* head_scan,
                                uint64_t* group_start, uint64_t* n_groups) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    bool head = (i == 0) || !mrec_eq(recs[i], recs[i - 1]);
    if (head) group_start[head_scan[i]] = i;
    // total groups = exclusive-scan of flags at the last element PLUS its own
    // flag (the last record need not start a group)
    if (i == n - 1) *n_groups = head_scan[i] + (head ? 1 : 0);
}

__global__ void k_init_outstats(OutStats* st) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        memset(st, 0, sizeof(OutStats));
        st->min_ts_flip = 0xFFFFFFFFFFFFFFFFULL;
        st->min_ldt_flip = 0xFFFFFFFFFFFFFFFFULL;
        st->min_ttl = 0xFFFFFFFFu;
        st->first_group = 0xFFFFFFFFFFFFFFFFULL;
    }
}

}  // namespace gpuc

namespace gpuc {

// ---------------------------------------------------------------------------
// serialized partition size / emit (simple schema; mirrors the oracle spec of
// UnfilteredSerializer.java:151-210 + Cell.java:268-306 +
// SortedTablePartitionWriter.java:97-166 byte for byte)
// ---------------------------------------------------------------------------
struct SerParams {
    HeaderStats hs;
    int32_t col_fixed_len;  // -1 = variable (vint length prefix)
};

__device__ inline uint32_t d_row_body_size(const OutParts& op, uint64_t g, const SerParams& sp,
                                           uint8_t* out_rflags, uint8_t* out_cflags) {
    uint8_t f = op.flags[g];
    uint32_t body = 0;
    uint8_t rflags = 0;
    int64_t lts = op.live_ts[g];
    int32_t lttl = op.live_ttl[g];
    int64_t llet = op.live_let[g];
    bool live = f & PF_LIVE_TS;
    bool expiring_live = live && lttl != NO_TTL;
    bool rdel = f & PF_ROW_DEL;
    bool cell = f & PF_HAS_CELL;
    if (live) { rflags |= F_TS; body += uvint_size((uint64_t)(lts - sp.hs.min_ts)); }
    if (expiring_live) {
        rflags |= F_TTL;
        body += uvint_size(sext32(lttl - sp.hs.min_ttl));
        body += uvint_size(sext32(llet - sp.hs.min_ldt));
    }
    if (rdel) {
        rflags |= F_DEL;
        body += uvint_size((uint64_t)(op.rdel_mfda[g] - sp.hs.min_ts));
        body += uvint_size(sext32(ldt_long(op.rdel_ldt[g]) - sp.hs.min_ldt));
    }
    if (cell) rflags |= F_ALLCOL;
    else body += uvint_size(1);  // bitmap: the single column is missing
    uint8_t cflags = 0;
    if (cell) {
        int64_t cts = op.cell_ts[g];
        uint32_t cldt = op.cell_ldt[g];
        int32_t cttl = op.cell_ttl[g];
        uint32_t vlen = op.val_len[g];
        bool has_value = vlen > 0 && (f & PF_CELL_VALUE);
        bool deleted = cldt != LDT_NONE_U32 && cttl == NO_TTL;
        bool expiring = cttl != NO_TTL;
        bool use_row_ts = live && cts == lts;
        bool use_row_ttl = expiring && expiring_live && cttl == lttl && ldt_long(cldt) == llet;
        if (!has_value) cflags |= CF_EMPTY;
        if (deleted) cflags |= CF_DELETED;
        else if (expiring) cflags |= CF_EXPIRING;
        if (use_row_ts) cflags |= CF_ROWTS;
        if (use_row_ttl) cflags |= CF_ROWTTL;
        body += 1;
        if (!use_row_ts) body += uvint_size((uint64_t)(cts - sp.hs.min_ts));
        if ((deleted || expiring) && !use_row_ttl) body += uvint_size(sext32(ldt_long(cldt) - sp.hs.min_ldt));
        if (expiring && !use_row_ttl) body += uvint_size(sext32(cttl - sp.hs.min_ttl));
        if (has_value) body += (sp.col_fixed_len >= 0 ? 0 : uvint_size(vlen)) + vlen;
    }
    if (out_rflags) *out_rflags = rflags;
    if (out_cflags) *out_cflags = cflags;
    return body;
}

__device__ inline uint64_t d_partition_size(const OutParts& op, uint64_t g, const SerParams& sp) {
    if (!op.keep[g]) return 0;
    uint32_t klen = op.klen[g];
    bool pdel_live = op.pdel_mfda[g] == INT64_MIN && op.pdel_ldt[g] == LDT_NONE_U32;
    uint64_t header_len = 2 + klen + (pdel_live ? 1 : 12);
    uint64_t sz = header_len;
    if (op.flags[g] & PF_HAS_ROW) {
        uint32_t body = d_row_body_size(op, g, sp, nullptr, nullptr);
        uint32_t prev_vs = uvint_size(header_len);  // previousUnfilteredSize = header bytes
        sz += 1 + uvint_size(body + prev_vs) + prev_vs + body;
    }
    return sz + 1;  // END_OF_PARTITION
}

__global__ void k_sizes(OutParts op, uint64_t n, SerParams sp, uint64_t* psize,
                        OutStats* st, const int64_t* ps_hist_off, int32_t ps_hist_n,
                        const int64_t* ch_hist_off, int32_t ch_hist_n) {
    __shared__ unsigned int sh_ps[156], sh_ch[119];
    for (int i = threadIdx.x; i < 156; i += blockDim.x) sh_ps[i] = 0;
    for (int i = threadIdx.x; i < 119; i += blockDim.x) sh_ch[i] = 0;
    __syncthreads();
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g < n) {
        uint64_t s = d_partition_size(op, g, sp);
        psize[g] = s;
        if (s) {
            // EstimatedHistogram.add: first offset >= value
            int lo = 0, hi = ps_hist_n;
            while (lo < hi) { int mid = (lo + hi) >> 1; if ((uint64_t)ps_hist_off[mid] < s) lo = mid + 1; else hi = mid; }
            atomicAdd(&sh_ps[lo], 1u);
            uint64_t cells = (op.flags[g] & PF_HAS_CELL) ? 1 : 0;
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

__global__ void k_index_sizes(OutParts op, uint64_t n, const uint64_t* data_off,
                              uint64_t* isize) {
    uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n) return;
    // RowIndexEntry.serialize non-indexed: vint position + vint 0 (RowIndexEntry.java:468-473)
    isize[g] = op.keep[g] ? 2 + op.klen[g] + uvint_size(data_off[g]) + 1 : 0;
}

__global__ void k_serialize(OutParts op, uint64_t n, SerParams sp, const uint64_t* data_off,
                            const uint64_t* idx_off, uint8_t* out_data, uint8_t* out_index,
                            uint32_t* bloom_bits, uint64_t bloom_bitlen, int32_t bloom_k) {
    uint64_t g = blockIdx.x * (blockDim.x / WAVE) + (threadIdx.x / WAVE);
    if (g >= n || !op.keep[g]) return;
    int lane = threadIdx.x & (WAVE - 1);
    uint32_t klen = op.klen[g];
    uint64_t kp = op.keypfx[g];
    bool pdel_live = op.pdel_mfda[g] == INT64_MIN && op.pdel_ldt[g] == LDT_NONE_U32;
    uint64_t header_len = 2 + klen + (pdel_live ? 1 : 12);
    bool has_row = op.flags[g] & PF_HAS_ROW;
    uint8_t rflags = 0, cflags = 0;
    uint32_t body = has_row ? d_row_body_size(op, g, sp, &rflags, &cflags) : 0;

    // value copy (all lanes): destination = end of partition bytes - vlen - 1(end byte)
    uint32_t vlen = 0;
    uint64_t vdst = 0, vsrc = 0;
    if (has_row && (cflags & CF_EMPTY) == 0 && (op.flags[g] & PF_HAS_CELL)) {
        vlen = op.val_len[g];
        uint32_t prev_vs = uvint_size(header_len);
        uint64_t psz = header_len + 1 + uvint_size(body + prev_vs) + prev_vs + body + 1;
        vdst = data_off[g] + psz - 1 - vlen;
        vsrc = op.val_addr[g];
    }
    if (vlen) {
        for (uint32_t i = lane; i < vlen; i += WAVE) out_data[vdst + i] = ((const uint8_t*)vsrc)[i];
    }
    if (lane != 0) return;

    // ---- partition meta bytes ----
    uint8_t* p = out_data + data_off[g];
    *p++ = (uint8_t)(klen >> 8);
    *p++ = (uint8_t)klen;
    uint8_t keyb[8];
    for (uint32_t b = 0; b < klen; b++) { keyb[b] = (uint8_t)(kp >> (8 * (7 - b))); *p++ = keyb[b]; }
    if (pdel_live) *p++ = 0x80;
    else {
        uint64_t m = (uint64_t)op.pdel_mfda[g];
        for (int i = 0; i < 8; i++) *p++ = (uint8_t)(m >> (8 * (7 - i)));
        uint32_t l = op.pdel_ldt[g];
        for (int i = 0; i < 4; i++) *p++ = (uint8_t)(l >> (8 * (3 - i)));
    }
    if (has_row) {
        *p++ = rflags;
        uint32_t prev_vs = uvint_size(header_len);
        p += uvint_put(p, body + prev_vs);
        p += uvint_put(p, header_len);
        if (rflags & F_TS) p += uvint_put(p, (uint64_t)(op.live_ts[g] - sp.hs.min_ts));
        if (rflags & F_TTL) {
            p += uvint_put(p, sext32(op.live_ttl[g] - sp.hs.min_ttl));
            p += uvint_put(p, sext32(op.live_let[g] - sp.hs.min_ldt));
        }
        if (rflags & F_DEL) {
            p += uvint_put(p, (uint64_t)(op.rdel_mfda[g] - sp.hs.min_ts));
            p += uvint_put(p, sext32(ldt_long(op.rdel_ldt[g]) - sp.hs.min_ldt));
        }
        if (!(rflags & F_ALLCOL)) p += uvint_put(p, 1);
        if (op.flags[g] & PF_HAS_CELL) {
            *p++ = cflags;
            if (!(cflags & CF_ROWTS)) p += uvint_put(p, (uint64_t)(op.cell_ts[g] - sp.hs.min_ts));
            bool deleted = cflags & CF_DELETED, expiring = cflags & CF_EXPIRING;
            if ((deleted || expiring) && !(cflags & CF_ROWTTL))
                p += uvint_put(p, sext32(ldt_long(op.cell_ldt[g]) - sp.hs.min_ldt));
            if (expiring && !(cflags & CF_ROWTTL))
                p += uvint_put(p, sext32(op.cell_ttl[g] - sp.hs.min_ttl));
            if (!(cflags & CF_EMPTY)) {
                if (sp.col_fixed_len < 0) p += uvint_put(p, op.val_len[g]);
                p += op.val_len[g];  // value bytes written by all lanes above
            }
        }
    }
    *p++ = F_END;

    // ---- Index.db entry (key + vint position + vint 0) ----
    uint8_t* q = out_index + idx_off[g];
    *q++ = (uint8_t)(klen >> 8);
    *q++ = (uint8_t)klen;
    for (uint32_t b = 0; b < klen; b++) *q++ = keyb[b];
    q += uvint_put(q, data_off[g]);
    *q++ = 0;

    // ---- bloom (BloomFilter.java:104-122: base=h[1], inc=h[0], abs(base % max)) ----
    uint64_t h[2];
    murmur3_128(keyb, klen, 0, h);
    int64_t base = (int64_t)h[1], inc = (int64_t)h[0];
    for (int i = 0; i < bloom_k; i++) {
        int64_t m = base % (int64_t)bloom_bitlen;
        uint64_t idx = (uint64_t)((m ^ (m >> 63)) - (m >> 63));
        atomicOr(&bloom_bits[idx >> 5], 1u << (idx & 31));
        base += inc;
    }
}

// ---------------------------------------------------------------------------
// LZ4 chunk compress (one wave per chunk; lane0 drives the bit-exact model;
// lanes stage the chunk into LDS first)
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(WAVE) k_lz4_compress(const uint8_t* data, uint64_t data_len,
                                                       uint8_t* slots, uint32_t* csize,
                                                       uint32_t* ccrc, uint32_t n_chunks) {
    __shared__ uint8_t s_chunk[CHUNK_LEN];
    __shared__ uint16_t s_table[LZ4M_HASHTABLESIZE_U16];
    uint32_t c = blockIdx.x;
    if (c >= n_chunks) return;
    int lane = threadIdx.x;
    uint64_t off = (uint64_t)c * CHUNK_LEN;
    uint32_t len = (uint32_t)min((uint64_t)CHUNK_LEN, data_len - off);
    for (uint32_t i = lane * 4; i < len; i += WAVE * 4) {
        // 4-byte staging copies (tail handled bytewise)
        if (i + 4 <= len && ((off + i) & 3) == 0) *(uint32_t*)&s_chunk[i] = *(const uint32_t*)&data[off + i];
        else for (uint32_t b = i; b < min(i + 4u, len); b++) s_chunk[b] = data[off + b];
    }
    for (int i = lane; i < LZ4M_HASHTABLESIZE_U16; i += WAVE) s_table[i] = 0;
    __syncthreads();
    if (lane != 0) return;
    uint8_t* dst = slots + (uint64_t)c * LZ4_SLOT;
    // 4-byte LITTLE-endian uncompressed length header (LZ4Compressor.java:118-124)
    dst[0] = (uint8_t)len; dst[1] = (uint8_t)(len >> 8); dst[2] = (uint8_t)(len >> 16); dst[3] = (uint8_t)(len >> 24);
    int csz = lz4m_compress(s_chunk, (int)len, dst + 4, s_table);
    uint32_t total = (uint32_t)csz + 4;
    csize[c] = total;
    ccrc[c] = d_crc32(dst, total);  // CRC32 of the compressed bytes incl. header
}

// gather compressed chunks into the final contiguous Data.db image
__global__ void k_chunk_gather(const uint8_t* slots, const uint32_t* csize, const uint32_t* ccrc,
                               const uint64_t* foff, uint8_t* out, uint32_t n_chunks) {
    uint32_t c = blockIdx.x;
    if (c >= n_chunks) return;
    int lane = threadIdx.x;
    uint32_t sz = csize[c];
    const uint8_t* src = slots + (uint64_t)c * LZ4_SLOT;
    uint8_t* dst = out + foff[c] + (uint64_t)c * 4;  // +4 per preceding chunk CRC
    for (uint32_t i = lane; i < sz; i += WAVE) dst[i] = src[i];
    if (lane == 0) {
        uint32_t crc = ccrc[c];
        dst[sz] = (uint8_t)(crc >> 24); dst[sz + 1] = (uint8_t)(crc >> 16);
        dst[sz + 2] = (uint8_t)(crc >> 8); dst[sz + 3] = (uint8_t)crc;
    }
}

// ---------------------------------------------------------------------------
// synthetic generator kernels (shared contract with oracle/src/gen.h)
// ---------------------------------------------------------------------------
struct GenParams {
    uint64_t seed, universe, stride, rows;
    uint32_t sst;  // sstable index
    uint32_t value_len, value_repeat_pct, tombstone_pct, partition_del_pct;
    int64_t base_ts, base_ldt;
};

__global__ void k_gen_recs(GenParams gp, MRec* recs, uint64_t* ids) {
    uint64_t j = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (j >= gp.rows) return;
    uint64_t id = feistel_perm(gp.seed, gp.universe, (gp.sst * gp.stride + j) % gp.universe);
    ids[j] = id;
    uint8_t key[8];
    for (int b = 0; b < 8; b++) key[b] = (uint8_t)(id >> (8 * (7 - b)));
    int64_t tok = murmur3_token(key, 8);
    MRec r;
    r.tok = (uint64_t)tok ^ 0x8000000000000000ULL;
    r.pfx = id;  // key bytes BE == the id value
    r.idx = (uint32_t)j;
    r.src = 0;
    r.klen = 8;
    r.pad = 0;
    recs[j] = r;
}

__global__ void k_gen_fill(GenParams gp, const MRec* sorted, const uint64_t* ids, uint64_t n,
                           OutParts op, uint8_t* values) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t id = ids[sorted[i].idx];
    op.keypfx[i] = sorted[i].pfx;
    op.klen[i] = 8;
    op.keep[i] = 1;
    int64_t ts = gp.base_ts + (int64_t)(splitmix64(gp.seed ^ id * 31 ^ ((uint64_t)gp.sst << 48)) % 1000000000ULL);
    bool pdel = gp.partition_del_pct && (splitmix64(gp.seed ^ 0xFEEDULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 < gp.partition_del_pct);
    bool tomb = !pdel && gp.tombstone_pct && (splitmix64(gp.seed ^ 0xDEADULL ^ id ^ ((uint64_t)gp.sst << 32)) % 100 < gp.tombstone_pct);
    int64_t pdm = INT64_MIN, rdm = INT64_MIN, lts = NO_TIMESTAMP;
    uint32_t pdl = LDT_NONE_U32, rdl = LDT_NONE_U32;
    uint8_t f = 0;
    uint64_t vaddr = 0;
    uint32_t vlen = 0;
    if (pdel) {
        pdm = ts;
        pdl = (uint32_t)(gp.base_ldt + (int64_t)(splitmix64(id ^ 0xDD) % 1000));
    } else if (tomb) {
        f = PF_HAS_ROW | PF_ROW_DEL;
        rdm = ts;
        rdl = (uint32_t)(gp.base_ldt + (int64_t)(splitmix64(id ^ 0xEE) % 1000));
    } else {
        f = PF_HAS_ROW | PF_LIVE_TS | PF_HAS_CELL | PF_CELL_VALUE;
        lts = ts;
        vlen = gp.value_len;
        vaddr = (uint64_t)(values + i * (uint64_t)gp.value_len);
    }
    op.pdel_mfda[i] = pdm;
    op.pdel_ldt[i] = pdl;
    op.flags[i] = f;
    op.live_ts[i] = lts;
    op.live_ttl[i] = 0;
    op.live_let[i] = NO_DELETION_TIME;
    op.rdel_mfda[i] = rdm;
    op.rdel_ldt[i] = rdl;
    op.cell_ts[i] = lts;
    op.cell_ldt[i] = LDT_NONE_U32;
    op.cell_ttl[i] = 0;
    op.val_addr[i] = vaddr;
    op.val_len[i] = vlen;
}

__global__ void k_gen_values(GenParams gp, const MRec* sorted, const uint64_t* ids,
                             const OutParts op, uint64_t n, uint8_t* values) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (!(op.flags[i] & PF_CELL_VALUE)) return;
    uint64_t id = ids[sorted[i].idx];
    uint8_t* out = values + i * (uint64_t)gp.value_len;
    uint64_t state = gp.seed ^ id * 0x100000001B3ULL ^ ((uint64_t)gp.sst << 40);
    uint64_t prev = splitmix64(state);
    uint32_t nw = (gp.value_len + 7) / 8;
    for (uint32_t w = 0; w < nw; w++) {
        uint64_t r = splitmix64(state + 1 + w);
        uint64_t word = (r % 100 < gp.value_repeat_pct && w > 0) ? prev : splitmix64(r);
        prev = word;
        uint32_t off = w * 8;
        for (uint32_t b = 0; b < 8 && off + b < gp.value_len; b++)
            out[off + b] = (uint8_t)(word >> (8 * b));
    }
}

// small uniform runs (run_size < 16): one thread merges a whole pair
__global__ void k_merge_small(const MRec* in, MRec* out, uint64_t n, uint64_t run_size) {
    uint64_t p = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t a_beg = p * run_size * 2;
    if (a_beg >= n) return;
    uint64_t a_end = min(a_beg + run_size, n);
    uint64_t b_beg = a_end;
    uint64_t b_end = min(a_beg + run_size * 2, n);
    uint64_t i = a_beg, j = b_beg, o = a_beg;
    while (i < a_end && j < b_end)
        out[o++] = !mrec_less(in[j], in[i]) ? in[i++] : in[j++];
    while (i < a_end) out[o++] = in[i++];
    while (j < b_end) out[o++] = in[j++];
}

// uniform-run merge round for the generator sort (runs of size run_size >= 16)
__global__ void k_merge_uniform(const MRec* in, MRec* out, uint64_t n, uint64_t run_size) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t k0 = t * MERGE_TILE;
    if (k0 >= n) return;
    uint64_t pair_span = run_size * 2;
    uint64_t pair = k0 / pair_span;
    uint64_t a_beg = pair * pair_span;
    uint64_t a_end = min(a_beg + run_size, n);
    uint64_t b_beg = a_end;
    uint64_t b_end = min(a_beg + pair_span, n);
    uint64_t an = a_end - a_beg, bn = b_end - b_beg;
    uint64_t k = k0 - a_beg;                 // offset within pair
    uint64_t kend = min(k + (uint64_t)MERGE_TILE, an + bn);
    // clamp tile to this pair (MERGE_TILE divides pair_span when both are pow2)
    uint64_t ilo = k > bn ? k - bn : 0;
    uint64_t ihi = k < an ? k : an;
    while (ilo < ihi) {
        uint64_t i = (ilo + ihi + 1) >> 1;
        if (!mrec_less(in[b_beg + (k - i)], in[a_beg + i - 1])) ilo = i;
        else ihi = i - 1;
    }
    uint64_t i = ilo, j = k - ilo;
    for (uint64_t o = k; o < kend; o++) {
        bool take_a;
        if (i >= an) take_a = false;
        else if (j >= bn) take_a = true;
        else take_a = !mrec_less(in[b_beg + j], in[a_beg + i]);
        out[a_beg + o] = take_a ? in[a_beg + i++] : in[b_beg + j++];
    }
}

}  // namespace gpuc

#include "lz4_wave.h"
