// PRODUCT — device-side data structures for the compaction pipeline.
#pragma once
#include <stdint.h>

namespace gpuc {

// merge record: one input partition version. 24 bytes.
// Sort key = (tok, pfx, klen): tok is the Murmur3 token with the sign bit
// flipped so unsigned compare == signed token order (DecoratedKey order:
// token, then unsigned key bytes — DecoratedKey.java:79-92; key bytes ≤ 8,
// zero-padded big-endian in pfx, ties broken by klen).
struct MRec {
    uint64_t tok;   // token ^ 0x8000000000000000
    uint64_t pfx;   // first min(8,klen) key bytes, big-endian, zero-padded
    uint32_t idx;   // partition ordinal within source
    uint16_t src;
    uint16_t klen;  // full key length (Cassandra keys are u16-length)
};

// Key lookup for exact DecoratedKey comparison beyond the 8-byte prefix:
// base[src] + pos[src][idx] + 2 is the address of partition idx's key bytes
// in the decompressed input (the short-length field precedes them). The
// compaction path enables it after parse; the generator path disables it
// (generated keys have unique 8-byte prefixes, so prefix+klen is exact).
struct KeyLut {
    const uint8_t* base[64];
    const uint64_t* pos[64];
    int enabled;
};

__device__ inline const uint8_t* mrec_key_bytes(const KeyLut& lut, const MRec& r) {
    return lut.base[r.src] + lut.pos[r.src][r.idx] + 2;
}
// DecoratedKey order (DecoratedKey.java:79-92): token, then key bytes
// compared as unsigned lexicographic with shorter-is-less on prefix equality
// (ByteBufferUtil.compareUnsigned). The zero-padded 8-byte prefix compare is
// exact whenever it differs; ties fall back to a byte walk from offset 8.
__device__ inline bool mrec_less(const KeyLut& lut, const MRec& a, const MRec& b) {
    if (a.tok != b.tok) return a.tok < b.tok;
    if (a.pfx != b.pfx) return a.pfx < b.pfx;
    if (!lut.enabled || (a.klen <= 8 && b.klen <= 8)) return a.klen < b.klen;
    const uint8_t* ka = mrec_key_bytes(lut, a);
    const uint8_t* kb = mrec_key_bytes(lut, b);
    uint32_t n = a.klen < b.klen ? a.klen : b.klen;
    for (uint32_t i = 8; i < n; i++)
        if (ka[i] != kb[i]) return ka[i] < kb[i];
    return a.klen < b.klen;
}
__device__ inline bool mrec_eq(const KeyLut& lut, const MRec& a, const MRec& b) {
    if (a.tok != b.tok || a.pfx != b.pfx || a.klen != b.klen) return false;
    if (!lut.enabled || a.klen <= 8) return true;
    const uint8_t* ka = mrec_key_bytes(lut, a);
    const uint8_t* kb = mrec_key_bytes(lut, b);
    for (uint32_t i = 8; i < a.klen; i++)
        if (ka[i] != kb[i]) return false;
    return true;
}

// unfiltered-level SoA (rows AND range-tombstone markers), used for parsed
// input rows and for reconciled output rows. One clustering column of fixed
// width (LongType/Int32Type) or none; `ck` is the sortable encoding
// (big-endian fixed value as signed -> value ^ SIGN so unsigned compare ==
// the ClusteringComparator order). `rkind` is the ClusteringPrefix.Kind
// ordinal (CLUSTERING_K = 4 for rows; bound/boundary kinds for markers).
struct UnfCols {
    uint64_t* ck;
    uint8_t* rkind;
    uint8_t* flags;      // PF_* for rows
    int64_t* live_ts;
    int32_t* live_ttl;
    int64_t* live_let;
    int64_t* rdel_mfda;  // row deletion, or marker end/close deletion
    uint32_t* rdel_ldt;
    int64_t* start_mfda;  // marker open deletion (boundary markers)
    uint32_t* start_ldt;
    // regular-column cells, row-major strided: index = row_o * n_cols + c
    // (n_cols from SchemaParams; n_cols == 1 degenerates to the old layout)
    int64_t* cell_ts;
    uint32_t* cell_ldt;
    int32_t* cell_ttl;
    uint64_t* val_addr;   // absolute device address of value bytes
    uint32_t* val_len;
    uint8_t* cell_flags;  // CF_* per cell
    // clustering components, strided by n_ck: index = row_o * n_ck + c.
    // Fixed-width components store the sign-flipped big-endian value in `ck`;
    // variable-width components (UTF8/Bytes) store the first min(8,len) bytes
    // big-endian zero-padded (unsigned lex prefix, no flip) with the full
    // bytes at ck_addr/ck_len breaking prefix ties. ck_count = components
    // present (rows: n_ck; bounds may be shorter prefixes).
    uint64_t* ck_addr;
    uint32_t* ck_len;
    uint8_t* ck_count;
    // ONE complex column (MapType(BytesType,BytesType)), the LAST regular
    // column when SchemaParams.n_cpx == 1 (ComplexColumnData.java:47): a
    // per-row complexDeletion + a [cpx_start, cpx_start+cpx_count) segment
    // of path-ordered cells in the CpxCells arena. PF_HAS_CPX marks column
    // presence (deletion-only columns have cpx_count == 0).
    int64_t* cpx_del_mfda;
    uint32_t* cpx_del_ldt;
    uint64_t* cpx_start;
    uint32_t* cpx_count;
    struct CpxCells {
        int64_t* ts;
        uint32_t* ldt;
        int32_t* ttl;
        uint8_t* flags;       // CELLF_*
        uint64_t* path_addr;  // CellPath bytes (map key)
        uint32_t* path_len;
        uint64_t* val_addr;
        uint32_t* val_len;
    } cpx;
};
enum : uint8_t { PF_HAS_ROW = 1, PF_LIVE_TS = 2, PF_ROW_DEL = 4, PF_HAS_CPX = 8 };
// per-cell flags (cell_flags array)
enum : uint8_t { CELLF_PRESENT = 1, CELLF_HAS_VALUE = 2, CELLF_EXPIRING = 4 };
// ClusteringPrefix.Kind ordinals (ClusteringPrefix.java:65-85)
enum : uint8_t {
    BK_EXCL_END = 0, BK_INCL_START = 1, BK_EXCL_END_INCL_START = 2, BK_STATIC = 3,
    BK_CLUSTERING = 4, BK_INCL_END_EXCL_START = 5, BK_INCL_END = 6, BK_EXCL_START = 7
};
__host__ __device__ inline int bk_comparison(uint8_t k) {
    // Kind(comparison, …) ctor args (ClusteringPrefix.java:70-79)
    const int tbl[8] = {0, 0, 0, 1, 2, 3, 3, 3};
    return tbl[k];
}
__host__ __device__ inline bool bk_is_boundary(uint8_t k) {
    return k == BK_EXCL_END_INCL_START || k == BK_INCL_END_EXCL_START;
}
__host__ __device__ inline bool bk_is_open(uint8_t k) {
    return k == BK_INCL_START || k == BK_EXCL_START || bk_is_boundary(k);
}

// one static row, stored at PARTITION level (SortedTablePartitionWriter
// writes it between the partition deletion and the unfiltereds whenever the
// header has static columns; it never interacts with the marker machinery).
// Cells strided by n_static: index = part * n_static + c.
struct StaticCols {
    uint8_t* flags;       // PF_HAS_ROW | PF_LIVE_TS | PF_ROW_DEL (0 = empty row)
    int64_t* live_ts;
    int32_t* live_ttl;
    int64_t* live_let;
    int64_t* rdel_mfda;
    uint32_t* rdel_ldt;
    uint8_t* cell_flags;  // CELLF_* per static cell
    int64_t* cell_ts;
    uint32_t* cell_ldt;
    int32_t* cell_ttl;
    uint64_t* val_addr;
    uint32_t* val_len;
};

// parsed per input-partition fields (partition level)
struct ParsedCols {
    int64_t* pdel_mfda;
    uint32_t* pdel_ldt;
    uint32_t* row_count;  // unfiltereds in this partition
    uint64_t* row_base;   // start into the per-source-concatenated UnfCols
    uint64_t* key_addr;   // device address of the key bytes in input data
    uint32_t* cpx_total;  // complex cells in this partition (arena sizing pass)
    StaticCols st;        // static row per partition (when the schema has one)
};

// reconciled output partitions (also produced directly by the generator)
struct OutParts {
    uint64_t* keypfx;    // big-endian zero-padded key prefix (debug/aux)
    uint64_t* key_addr;  // device address of the full key bytes
    int64_t* token;      // Murmur3 token (garbage-collect source matching)
    StaticCols st;       // merged static row per output partition
    uint16_t* klen;
    int64_t* pdel_mfda;
    uint32_t* pdel_ldt;
    uint64_t* row_base;  // into the output UnfCols arena
    uint32_t* row_count; // merged+purged unfiltereds
    uint8_t* keep;       // 0 = dropped (purged empty / out of shard)
    uint8_t* merged_k;   // versions merged per group (histogram fed by k_merged_hist)
};

// EncodingStats bases of the OUTPUT header (SerializationHeader.make)
struct HeaderStats {
    int64_t min_ts;
    int64_t min_ldt;
    int32_t min_ttl;
};

// device-collected output metadata (atomics)
struct OutStats {
    unsigned long long partitions_out;
    unsigned long long rows_out;
    unsigned long long total_cells;
    unsigned long long min_ts_flip, max_ts_flip;     // i64 ^ SIGN, min/max as u64
    unsigned long long min_ldt_flip, max_ldt_flip;
    unsigned int min_ttl, max_ttl;                   // ttl >= 0
    unsigned long long has_partition_deletions;
    unsigned long long first_group, last_group;      // min/max kept group index
    unsigned long long merged_counts[64];
    unsigned long long part_size_hist[156];          // EstimatedHistogram(155)
    unsigned long long cells_hist[119];              // EstimatedHistogram(118)
    unsigned long long tomb_count;                   // appended ldt list counter
    unsigned long long error;                        // parse/feature error flag
    unsigned long long cpx_alloc;                    // output cpx-cell arena bump
};

constexpr uint32_t LZ4_SLOT = 16480;  // > 4 + LZ4_compressBound(16384), 16B aligned
constexpr uint32_t SNP_SLOT = 19456;  // > snappy_max_compressed_length(16384)=19146, 16B aligned

}  // namespace gpuc
