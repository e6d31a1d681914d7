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

// ---------------------------------------------------------------------------
// pairwise stable merge (merge path, per-thread tiles)
// ---------------------------------------------------------------------------
struct MergePair {
    uint64_t a_beg, a_end, b_beg, b_end, out_beg;
    uint64_t tile_beg;  // first global tile of this pair
};

#define MERGE_TILE 16

__global__ void k_merge_pairs(const MRec* in, MRec* out, const MergePair* pairs,
                              uint32_t n_pairs, uint64_t total_tiles, KeyLut lut) {
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
        if (!mrec_less(lut, in[p.b_beg + (k - i)], in[p.a_beg + i - 1])) ilo = i;
        else ihi = i - 1;
    }
    uint64_t i = ilo, j = k - ilo;
    for (uint64_t o = k; o < kend; o++) {
        bool take_a;
        if (i >= an) take_a = false;
        else if (j >= bn) take_a = true;
        else take_a = !mrec_less(lut, in[p.b_beg + j], in[p.a_beg + i]);  // ties -> A
        out[p.out_beg + o] = take_a ? in[p.a_beg + i++] : in[p.b_beg + j++];
    }
}

// ---------------------------------------------------------------------------
// group heads + exclusive scan utilities (u64)
// ---------------------------------------------------------------------------
__global__ void k_group_heads(const MRec* recs, uint64_t n, uint64_t* head, KeyLut lut) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    head[i] = (i == 0) || !mrec_eq(lut, recs[i], recs[i - 1]);
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
// group starts using the scanned head values (heads recomputed from recs)
__global__ void k_group_starts2(const MRec* recs, uint64_t n, const uint64_t* head_scan,
                                uint64_t* group_start, uint64_t* n_groups, KeyLut lut) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    bool head = (i == 0) || !mrec_eq(lut, recs[i], recs[i - 1]);
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
// ---------------------------------------------------------------------------
// gather compressed chunks into the final contiguous Data.db image
__global__ void k_chunk_gather(const uint8_t* slots, const uint32_t* csize, const uint32_t* ccrc,
                               const uint64_t* foff, uint8_t* out, uint32_t n_chunks,
                               uint32_t slot_stride = LZ4_SLOT) {
    uint32_t c = blockIdx.x;
    if (c >= n_chunks) return;
    int lane = threadIdx.x;
    uint32_t sz = csize[c];
    const uint8_t* src = slots + (uint64_t)c * slot_stride;
    uint8_t* dst = out + foff[c] + (uint64_t)c * 4;  // +4 per preceding chunk CRC
    for (uint32_t i = 4u * lane; i < sz; i += 4u * WAVE) {
        uint32_t nb = sz - i;
        if (nb >= 4) {
            uint32_t v;
            memcpy(&v, src + i, 4);
            memcpy(dst + i, &v, 4);
        } else {
            for (uint32_t j = 0; j < nb; j++) dst[i + j] = src[i + j];
        }
    }
    if (lane == 0) {
        uint32_t crc = ccrc[c];
        dst[sz] = (uint8_t)(crc >> 24); dst[sz + 1] = (uint8_t)(crc >> 16);
        dst[sz + 2] = (uint8_t)(crc >> 8); dst[sz + 3] = (uint8_t)crc;
    }
}

// ---------------------------------------------------------------------------
// synthetic generator kernels (shared contract with oracle/src/gen.h)
// ---------------------------------------------------------------------------
// small uniform runs (run_size < 16): one thread merges a whole pair
__global__ void k_merge_small(const MRec* in, MRec* out, uint64_t n, uint64_t run_size, KeyLut lut) {
    uint64_t p = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t a_beg = p * run_size * 2;
    if (a_beg >= n) return;
    uint64_t a_end = min(a_beg + run_size, n);
    uint64_t b_beg = a_end;
    uint64_t b_end = min(a_beg + run_size * 2, n);
    uint64_t i = a_beg, j = b_beg, o = a_beg;
    while (i < a_end && j < b_end)
        out[o++] = !mrec_less(lut, in[j], in[i]) ? in[i++] : in[j++];
    while (i < a_end) out[o++] = in[i++];
    while (j < b_end) out[o++] = in[j++];
}

// uniform-run merge round for the generator sort (runs of size run_size >= 16)
__global__ void k_merge_uniform(const MRec* in, MRec* out, uint64_t n, uint64_t run_size, KeyLut lut) {
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
        if (!mrec_less(lut, in[b_beg + (k - i)], in[a_beg + i - 1])) ilo = i;
        else ihi = i - 1;
    }
    uint64_t i = ilo, j = k - ilo;
    for (uint64_t o = k; o < kend; o++) {
        bool take_a;
        if (i >= an) take_a = false;
        else if (j >= bn) take_a = true;
        else take_a = !mrec_less(lut, in[b_beg + j], in[a_beg + i]);
        out[a_beg + o] = take_a ? in[a_beg + i++] : in[b_beg + j++];
    }
}

}  // namespace gpuc

#include "lz4_wave.h"
#include "snappy_wave.h"
