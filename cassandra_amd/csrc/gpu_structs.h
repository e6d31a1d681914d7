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
    uint64_t pfx;   // key bytes, big-endian, zero-padded
    uint32_t idx;   // partition ordinal within source
    uint16_t src;
    uint8_t klen;
    uint8_t pad;
};

__host__ __device__ inline bool mrec_less(const MRec& a, const MRec& b) {
    if (a.tok != b.tok) return a.tok < b.tok;
    if (a.pfx != b.pfx) return a.pfx < b.pfx;
    return a.klen < b.klen;
}
__host__ __device__ inline bool mrec_eq(const MRec& a, const MRec& b) {
    return a.tok == b.tok && a.pfx == b.pfx && a.klen == b.klen;
}

// parsed per input-partition fields (SoA, concatenated across sources; index
// space == MRec (src, idx) resolved through per-source base offsets)
struct ParsedCols {
    int64_t* pdel_mfda;
    uint32_t* pdel_ldt;
    uint8_t* flags;       // bit0 has_row, bit1 row_live_ts, bit2 row_del,
                          // bit3 has_cell, bit4 cell_has_value, bit5 cell_expiring
    int64_t* live_ts;
    int32_t* live_ttl;
    int64_t* live_let;
    int64_t* rdel_mfda;
    uint32_t* rdel_ldt;
    int64_t* cell_ts;
    uint32_t* cell_ldt;
    int32_t* cell_ttl;
    uint64_t* val_addr;   // absolute device address of value bytes
    uint32_t* val_len;
};
enum : uint8_t {
    PF_HAS_ROW = 1, PF_LIVE_TS = 2, PF_ROW_DEL = 4, PF_HAS_CELL = 8,
    PF_CELL_VALUE = 16, PF_CELL_EXPIRING = 32
};

// reconciled output partitions (also produced directly by the generator)
struct OutParts {
    uint64_t* keypfx;    // big-endian zero-padded key
    uint8_t* klen;
    int64_t* pdel_mfda;
    uint32_t* pdel_ldt;
    uint8_t* flags;      // PF_* as above; PF_HAS_ROW==0 -> deletion-only partition
    int64_t* live_ts;
    int32_t* live_ttl;
    int64_t* live_let;
    int64_t* rdel_mfda;
    uint32_t* rdel_ldt;
    int64_t* cell_ts;
    uint32_t* cell_ldt;
    int32_t* cell_ttl;
    uint64_t* val_addr;
    uint32_t* val_len;
    uint8_t* keep;       // 0 = dropped (purged empty / out of shard)
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
};

constexpr uint32_t LZ4_SLOT = 16480;  // > 4 + LZ4_compressBound(16384), 16B aligned

}  // namespace gpuc
