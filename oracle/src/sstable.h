// ORACLE — test infrastructure only (see util.h header note).
// In-memory model + reader/writer for Cassandra big-format `oa` sstables.
// Every encoding follows the reference file:line cited at the function.
#pragma once
#include "util.h"
#include <optional>
#include <map>

namespace oracle {

// ---- constants (db/LivenessInfo.java:42-51, db/rows/Cell.java:48-57,
//      db/rows/EncodingStats.java:45-67) ----
constexpr int64_t NO_TIMESTAMP = INT64_MIN;
constexpr int32_t NO_TTL = 0;
constexpr int32_t EXPIRED_LIVENESS_TTL = INT32_MAX;
constexpr int64_t NO_DELETION_TIME = INT64_MAX;       // long-semantics "none"
constexpr uint32_t LDT_NONE_U32 = 0xFFFFFFFFu;        // CassandraUInt encoding of "none"
constexpr int64_t NO_EXPIRATION_TIME = INT64_MAX;
constexpr int64_t TIMESTAMP_EPOCH = 1442880000000000LL;  // 2015-09-22T00:00:00Z µs
constexpr int64_t DELETION_TIME_EPOCH = 1442880000LL;    // seconds
constexpr uint32_t DEFAULT_CHUNK_LEN = 16384;            // schema/CompressionParams.java:47
constexpr uint32_t DEFAULT_MAX_COMPRESSED = 0x7FFFFFFF;  // min_compress_ratio=0 default
constexpr int COLUMN_INDEX_SIZE = 64 * 1024;             // Config.java:331 / BigFormatPartitionWriter.DEFAULT_GRANULARITY

// ldt long <-> u32 (Cell.java:81-90)
inline int64_t ldt_to_long(uint32_t u) { return u == LDT_NONE_U32 ? NO_DELETION_TIME : (int64_t)u; }
inline uint32_t ldt_to_u32(int64_t l) { return l == NO_DELETION_TIME ? LDT_NONE_U32 : (uint32_t)l; }

// ---- types (db/marshal/*) ----
enum class CqlType : uint8_t { BYTES, UTF8, ASCII, LONG, INT32, MAP_BB, COUNTER };
// CounterColumnType cells: value = a counter CONTEXT (CounterContext.java:
// header of global/local flags + (CounterId, clock, count) shards); variable
// width on the wire like BytesType, but reconciled by context merge.
inline bool is_counter_type(CqlType t) { return t == CqlType::COUNTER; }
// MAP_BB == MapType(BytesType,BytesType): the one COMPLEX column type the
// engine carries (ColumnMetadata.isComplex; db/rows/ComplexColumnData.java:47).
// Cell paths are the map keys, compared as BytesType (unsigned lexicographic).
inline bool is_complex_type(CqlType t) { return t == CqlType::MAP_BB; }
const char* cql_type_name(CqlType t);
CqlType cql_type_from_name(const std::string& java_name);
inline int fixed_len(CqlType t) {
    switch (t) { case CqlType::LONG: return 8; case CqlType::INT32: return 4; default: return -1; }
}
// value compare, per AbstractType: LongType/Int32Type signed numeric; UTF8/ASCII/Bytes unsigned lexicographic
int compare_typed(CqlType t, const bytes& a, const bytes& b);

// ---- DeletionTime (db/DeletionTime.java) ----
struct DeletionTime {
    int64_t mfda = INT64_MIN;          // markedForDeleteAt
    uint32_t ldt = LDT_NONE_U32;       // localDeletionTime, unsigned-int encoding
    bool live() const { return mfda == INT64_MIN && ldt == LDT_NONE_U32; }
    // DeletionTime.supersedes (DeletionTime.java:158-161)
    bool supersedes(const DeletionTime& o) const {
        return mfda > o.mfda || (mfda == o.mfda && ldt_to_long(ldt) > ldt_to_long(o.ldt));
    }
    // DeletionTime.deletes(ts) (DeletionTime.java:173-176)
    bool deletes(int64_t ts) const { return ts <= mfda; }
    bool operator==(const DeletionTime& o) const { return mfda == o.mfda && ldt == o.ldt; }
};
inline const DeletionTime DT_LIVE{};

// ---- LivenessInfo (db/LivenessInfo.java) ----
struct LivenessInfo {
    int64_t ts = NO_TIMESTAMP;
    int32_t ttl = NO_TTL;
    int64_t let = NO_EXPIRATION_TIME;  // localExpirationTime (long semantics)
    bool empty() const { return ts == NO_TIMESTAMP; }
    bool expiring() const { return ttl != NO_TTL; }
    bool expired() const { return ttl == EXPIRED_LIVENESS_TTL; }  // ExpiredLivenessInfo
    // LivenessInfo.supersedes (LivenessInfo.java:~190)
    bool supersedes(const LivenessInfo& o) const {
        if (ts != o.ts) return ts > o.ts;
        if (expired() != o.expired()) return expired();
        if (expiring() == o.expiring()) return let > o.let;
        return expiring();
    }
};

// ---- Cell (db/rows/Cell.java, AbstractCell) ----
struct Cell {
    int64_t ts = NO_TIMESTAMP;
    uint32_t ldt = LDT_NONE_U32;  // localDeletionTime (u32 encoding)
    int32_t ttl = NO_TTL;
    bytes value;                  // empty + !has_value -> HAS_EMPTY_VALUE
    bytes path;                   // CellPath (complex columns only): map key
    bool tombstone() const { return ldt != LDT_NONE_U32 && ttl == NO_TTL; }
    bool expiring() const { return ttl != NO_TTL; }
    // AbstractCell.isLive(nowInSec)
    bool is_live(int64_t now) const {
        return ldt == LDT_NONE_U32 || (ttl != NO_TTL && now < ldt_to_long(ldt));
    }
};

// ---- clustering values ----
struct ClusterVal {
    enum State : uint8_t { VALUE = 0, EMPTY = 1, NUL = 2 } state = VALUE;
    bytes v;
};
using Clustering = std::vector<ClusterVal>;
// comparator over clustering values of the same prefix length handled in compact.cpp

// ---- Unfiltered: row or range-tombstone marker ----
// kind ordinals per ClusteringPrefix.Kind (ClusteringPrefix.java:65-85)
enum BoundKind : uint8_t {
    EXCL_END = 0, INCL_START = 1, EXCL_END_INCL_START = 2, STATIC_K = 3,
    CLUSTERING_K = 4, INCL_END_EXCL_START = 5, INCL_END = 6, EXCL_START = 7
};

// CellPath comparator: map keys compare as the key type (BytesType here)
inline int compare_cell_path(const bytes& a, const bytes& b) {
    size_t n = a.size() < b.size() ? a.size() : b.size();
    int c = n ? __builtin_memcmp(a.data(), b.data(), n) : 0;
    if (c) return c;
    return a.size() == b.size() ? 0 : (a.size() < b.size() ? -1 : 1);
}

// db/rows/ComplexColumnData.java:47 — a complex column's deletion + its
// path-ordered cells. A ComplexData with live deletion and no cells is
// represented as absent (ComplexColumnData.update returns null then).
struct ComplexData {
    DeletionTime del;          // complexDeletion
    std::vector<Cell> cells;   // ordered by compare_cell_path
};

struct Row;
inline bool row_is_empty(const Row& r);
struct Row {
    Clustering clustering;
    LivenessInfo live;
    DeletionTime del;
    bool static_flag = false;
    std::vector<std::optional<Cell>> cells;  // index == header regular-column index
    // complex column data, same indexing (empty vector when the schema has
    // no complex columns; entries at simple-column indices stay nullopt)
    std::vector<std::optional<ComplexData>> complex;
    bool col_present(size_t i) const {
        if (i < cells.size() && cells[i]) return true;
        return i < complex.size() && complex[i].has_value();
    }
    bool empty_row() const {
        if (!live.empty() || !del.live()) return false;
        for (auto& c : cells) if (c) return false;
        for (auto& c : complex) if (c) return false;
        return true;
    }
};
inline bool row_is_empty(const Row& r) {
    if (!r.live.empty() || !r.del.live()) return false;
    for (auto& c : r.cells)
        if (c) return false;
    for (auto& c : r.complex)
        if (c) return false;
    return true;
}

struct Marker {
    BoundKind kind;
    Clustering values;  // size <= clustering column count
    DeletionTime end_dt;    // close-deletion (boundary) or the single bound deletion
    DeletionTime start_dt;  // open-deletion (boundary only)
    bool boundary() const { return kind == EXCL_END_INCL_START || kind == INCL_END_EXCL_START; }
    bool open(bool) const { return kind == INCL_START || kind == EXCL_START || boundary(); }
    bool close(bool) const { return kind == INCL_END || kind == EXCL_END || boundary(); }
    DeletionTime open_dt() const { return boundary() ? start_dt : end_dt; }
    DeletionTime close_dt() const { return end_dt; }
};

struct Unfiltered {
    enum Kind : uint8_t { ROW, MARKER } kind = ROW;
    Row row;
    Marker marker;
    const Clustering& clustering() const { return kind == ROW ? row.clustering : marker.values; }
};

struct Partition {
    bytes key;
    int64_t token = 0;
    DeletionTime del;
    Row static_row;   // static_flag=true when header.has_static(); may be empty
    std::vector<Unfiltered> items;
    void set_token() { token = murmur3_token(key.data(), key.size()); }
};

// ---- EncodingStats (db/rows/EncodingStats.java) ----
struct EncodingStats {
    int64_t min_ts = TIMESTAMP_EPOCH;
    int64_t min_ldt = DELETION_TIME_EPOCH;  // seconds (long)
    int32_t min_ttl = 0;
};

// ---- SerializationHeader (HEADER component of Statistics.db) ----
struct Header {
    EncodingStats stats;
    CqlType key_type = CqlType::BYTES;
    std::vector<CqlType> clustering_types;
    std::vector<std::pair<bytes, CqlType>> static_cols;   // insertion order preserved
    std::vector<std::pair<bytes, CqlType>> regular_cols;
    bool has_static() const { return !static_cols.empty(); }
    // delta codecs (SerializationHeader.java:165-200)
    void w_ts(bytes& o, int64_t ts) const { put_unsigned_vint(o, (uint64_t)(ts - stats.min_ts)); }
    // writeUnsignedVInt32 takes (int)(diff) then sign-extends to long (VIntCoding.java:329-332)
    void w_ldt(bytes& o, int64_t l) const { put_unsigned_vint(o, (uint64_t)(int64_t)(int32_t)(l - stats.min_ldt)); }
    void w_ttl(bytes& o, int32_t t) const { put_unsigned_vint(o, (uint64_t)(int64_t)(int32_t)(t - stats.min_ttl)); }
    void w_dt(bytes& o, const DeletionTime& dt) const { w_ts(o, dt.mfda); w_ldt(o, ldt_to_long(dt.ldt)); }
    int64_t r_ts(Reader& r) const { return (int64_t)read_unsigned_vint(r) + stats.min_ts; }
    int64_t r_ldt(Reader& r) const { return (int64_t)(int32_t)(uint32_t)read_unsigned_vint(r) + stats.min_ldt; }
    int32_t r_ttl(Reader& r) const { return (int32_t)(uint32_t)read_unsigned_vint(r) + stats.min_ttl; }
    DeletionTime r_dt(Reader& r) const {
        int64_t m = r_ts(r);
        int64_t l = r_ldt(r);
        return DeletionTime{m, ldt_to_u32(l)};
    }
};

// ---- compression ----
enum class Compressor : uint8_t { LZ4, SNAPPY, NONE };
struct CompressionParams {
    Compressor algo = Compressor::LZ4;
    uint32_t chunk_len = DEFAULT_CHUNK_LEN;
    uint32_t max_compressed_len = DEFAULT_MAX_COMPRESSED;
};

// ---- STATS component fields we consume/produce ----
struct StatsMins {
    int64_t min_timestamp = INT64_MAX;
    int64_t max_timestamp = INT64_MIN;
    int64_t min_ldt = NO_DELETION_TIME;   // long semantics
    int64_t max_ldt = INT64_MIN;
    int32_t min_ttl = INT32_MAX;
    int32_t max_ttl = 0;
    int64_t total_rows = 0;
    int64_t total_columns_set = 0;
};

enum class Partitioner : uint8_t { MURMUR3, BYTE_ORDERED };

struct SSTable {
    Header header;
    Partitioner partitioner = Partitioner::MURMUR3;
    CompressionParams comp;
    uint32_t column_index_size = COLUMN_INDEX_SIZE;  // Config.column_index_size (test conf: 4KiB)
    std::vector<Partition> parts;
    StatsMins stats;        // from STATS component (reader) / collected (writer)
    uint64_t generation = 0;
    // raw images kept by the reader for byte-level comparisons in tests
    bytes raw_data_uncompressed;
    bytes raw_statistics;   // whole Statistics.db as read
    bool bti = false;       // `da` (Big Trie-Indexed) component set
};

// ---- reader ----
// base = path prefix like "/dir/oa-1-big" (components appended as "-Data.db"...)
// scrub: salvage partitions untouched by corrupt chunks, rewrite clean.
// Recovery granularity (this implementation's contract, mirrored by the GPU):
// a partition is dropped iff any 16 KiB chunk overlapping its byte range
// fails CRC/decode; the rest is rewritten with header mins from the input
// Statistics (as a 1-input compaction would take them).
struct ScrubResult { uint64_t kept = 0, dropped = 0; };
ScrubResult scrub_sstable(const std::string& inbase, const std::string& outbase);

SSTable read_sstable(const std::string& base, bool keep_raw = false);
// parse one partition at `offset` of a decompressed Data.db image; optionally
// records each unfiltered's absolute byte offset (test/BTI-index tooling)
Partition read_one_partition(const bytes& raw, uint64_t offset, const Header& h,
                             std::vector<uint64_t>* item_offsets = nullptr);

// ---- writer ----
struct WriterOut {
    bytes data_db, index_db, compression_info, filter, digest, statistics, summary, toc;
    bytes partitions_db, rows_db;  // BTI (`da`) index components
    bool bti = false;
    uint64_t uncompressed_data_len = 0;
    uint64_t partition_count = 0;
};
// Serializes with sstable.header (stats deltas), comp params; bloom fp 0.01.
// bti=true additionally assembles Partitions.db/Rows.db (BtiFormat.md) from
// the SAME column_index_size blocks the big promoted index uses
// (BtiFormatPartitionWriter.java:52 — rowIndexBlockSize == column_index_size)
// and write_components emits the `da` component set instead of Index/Summary.
WriterOut write_sstable(const SSTable& t, bool bti = false);
void write_components(const WriterOut& w, const std::string& base);
// memtable-dump interchange for gpuc_flush_table parity (format ours; see .cpp)
void write_memdump(const SSTable& t, const std::string& path);

// serialize one partition into `out` (Data.db stream) and append its Index.db
// entry to `index_out`. Exposed for round-trip tests.
void serialize_partition(const Partition& p, const Header& h, bytes& out, bytes& index_out,
                         uint32_t column_index_size = COLUMN_INDEX_SIZE);
struct IndexInfoC;  // promoted-index block (defined in sstable.cpp)

// LZ4 chunk framing: compress `raw` into chunks (Data.db bytes), offsets, etc.
struct ChunkedOut { bytes file; std::vector<uint64_t> offsets; };
ChunkedOut chunk_compress(const bytes& raw, const CompressionParams& cp);
bytes make_compression_info(const CompressionParams& cp, uint64_t data_len,
                            const std::vector<uint64_t>& offsets);

}  // namespace oracle
