// ORACLE — test infrastructure only (see util.h header note).
// Reader/writer for Cassandra big-format `oa` sstables; bit-exact restatement
// of the serializers cited per function.
#include "sstable.h"
#include "bti.h"
#include <set>
#include "lz4_ref.h"
#include "snappy_ref.h"
#include <algorithm>
#include <cassert>
#include <cmath>

namespace oracle {

// ---------------------------------------------------------------------------
// types
// ---------------------------------------------------------------------------
const char* cql_type_name(CqlType t) {
    switch (t) {
        case CqlType::BYTES: return "org.apache.cassandra.db.marshal.BytesType";
        case CqlType::UTF8: return "org.apache.cassandra.db.marshal.UTF8Type";
        case CqlType::ASCII: return "org.apache.cassandra.db.marshal.AsciiType";
        case CqlType::LONG: return "org.apache.cassandra.db.marshal.LongType";
        case CqlType::INT32: return "org.apache.cassandra.db.marshal.Int32Type";
        case CqlType::MAP_BB:
            return "org.apache.cassandra.db.marshal.MapType"
                   "(org.apache.cassandra.db.marshal.BytesType,"
                   "org.apache.cassandra.db.marshal.BytesType)";
        case CqlType::COUNTER: return "org.apache.cassandra.db.marshal.CounterColumnType";
    }
    return "?";
}
CqlType cql_type_from_name(const std::string& n) {
    for (CqlType t : {CqlType::BYTES, CqlType::UTF8, CqlType::ASCII, CqlType::LONG, CqlType::INT32,
                      CqlType::MAP_BB, CqlType::COUNTER})
        if (n == cql_type_name(t)) return t;
    throw std::runtime_error("oracle: unsupported AbstractType " + n);
}
int compare_typed(CqlType t, const bytes& a, const bytes& b) {
    if (t == CqlType::LONG || t == CqlType::INT32) {
        // LongType/Int32Type.compare: signed numeric over full-width big-endian
        // bytes. Equivalent to unsigned-BE compare with the sign bit flipped.
        if (a.size() != (size_t)fixed_len(t) || b.size() != (size_t)fixed_len(t))
            throw std::runtime_error("bad fixed-width clustering value");
        uint8_t a0 = a[0] ^ 0x80, b0 = b[0] ^ 0x80;
        if (a0 != b0) return a0 < b0 ? -1 : 1;
        int c = memcmp(a.data() + 1, b.data() + 1, a.size() - 1);
        return c < 0 ? -1 : c > 0 ? 1 : 0;
    }
    size_t n = std::min(a.size(), b.size());
    int c = memcmp(a.data(), b.data(), n);
    if (c) return c;
    return a.size() == b.size() ? 0 : (a.size() < b.size() ? -1 : 1);
}

// ---------------------------------------------------------------------------
// DeletionTime `oa` serializer (DeletionTime.java:205-260)
// ---------------------------------------------------------------------------
static void put_deletion_time(bytes& out, const DeletionTime& dt) {
    if (dt.live()) { out.push_back(0x80); return; }
    put_be64(out, (uint64_t)dt.mfda);
    put_be32(out, dt.ldt);
}
static DeletionTime read_deletion_time(Reader& r) {
    uint8_t flags = r.p[r.pos];
    if (flags & 0x80) {
        r.skip(1);
        if (flags != 0x80) throw std::runtime_error("corrupt DeletionTime flags");
        return DT_LIVE;
    }
    int64_t mfda = (int64_t)r.be64();
    uint32_t ldt = r.be32();
    return DeletionTime{mfda, ldt};
}

// ---------------------------------------------------------------------------
// clustering values (ClusteringPrefix.java:455-540)
// ---------------------------------------------------------------------------
static uint64_t clustering_header(const Clustering& c, size_t off, size_t lim) {
    uint64_t h = 0;
    for (size_t i = off; i < lim; i++) {
        if (c[i].state == ClusterVal::NUL) h |= 1ULL << (i * 2 + 1);
        else if (c[i].state == ClusterVal::EMPTY) h |= 1ULL << (i * 2);
    }
    return h;
}
static void put_typed_value(bytes& out, CqlType t, const bytes& v) {
    // AbstractType.writeValue (AbstractType.java:529-560): fixed width raw, else vint length
    int fl = fixed_len(t);
    if (fl >= 0) {
        if ((int)v.size() != fl) throw std::runtime_error("fixed-width value size mismatch");
        out.insert(out.end(), v.begin(), v.end());
    } else {
        put_unsigned_vint(out, v.size());
        out.insert(out.end(), v.begin(), v.end());
    }
}
static bytes read_typed_value(Reader& r, CqlType t) {
    int fl = fixed_len(t);
    size_t n = fl >= 0 ? (size_t)fl : (size_t)read_unsigned_vint(r);
    return r.take(n);
}
static void put_clustering_values(bytes& out, const Clustering& c,
                                  const std::vector<CqlType>& types) {
    size_t off = 0, n = c.size();
    while (off < n) {
        size_t lim = std::min(n, off + 32);
        put_unsigned_vint(out, clustering_header(c, off, lim));
        for (; off < lim; off++)
            if (c[off].state == ClusterVal::VALUE)
                put_typed_value(out, types[off], c[off].v);
    }
}
static Clustering read_clustering_values(Reader& r, size_t n, const std::vector<CqlType>& types) {
    Clustering c(n);
    size_t off = 0;
    while (off < n) {
        size_t lim = std::min(n, off + 32);
        uint64_t h = read_unsigned_vint(r);
        for (; off < lim; off++) {
            if (h & (1ULL << (off * 2 + 1))) c[off].state = ClusterVal::NUL;
            else if (h & (1ULL << (off * 2))) c[off].state = ClusterVal::EMPTY;
            else { c[off].state = ClusterVal::VALUE; c[off].v = read_typed_value(r, types[off]); }
        }
    }
    return c;
}

// ---------------------------------------------------------------------------
// Unfiltered flags (UnfilteredSerializer.java:102-122)
// ---------------------------------------------------------------------------
enum : uint8_t {
    F_END_OF_PARTITION = 0x01, F_IS_MARKER = 0x02, F_HAS_TIMESTAMP = 0x04,
    F_HAS_TTL = 0x08, F_HAS_DELETION = 0x10, F_HAS_ALL_COLUMNS = 0x20,
    F_HAS_COMPLEX_DELETION = 0x40, F_EXTENSION = 0x80
};
enum : uint8_t { XF_IS_STATIC = 0x01 };
// cell flags (Cell.java:261-266)
enum : uint8_t {
    CF_IS_DELETED = 0x01, CF_IS_EXPIRING = 0x02, CF_HAS_EMPTY_VALUE = 0x04,
    CF_USE_ROW_TIMESTAMP = 0x08, CF_USE_ROW_TTL = 0x10
};

// ---- cell serialize (Cell.java:268-306) ----
static void put_cell(bytes& out, const Cell& c, CqlType type, const LivenessInfo& row_live,
                     const Header& h, bool complex = false) {
    bool has_value = !c.value.empty();
    bool is_deleted = c.tombstone();
    bool is_expiring = c.expiring();
    bool use_row_ts = !row_live.empty() && c.ts == row_live.ts;
    bool use_row_ttl = is_expiring && row_live.expiring() && c.ttl == row_live.ttl
                       && ldt_to_long(c.ldt) == row_live.let;
    uint8_t flags = 0;
    if (!has_value) flags |= CF_HAS_EMPTY_VALUE;
    if (is_deleted) flags |= CF_IS_DELETED;
    else if (is_expiring) flags |= CF_IS_EXPIRING;
    if (use_row_ts) flags |= CF_USE_ROW_TIMESTAMP;
    if (use_row_ttl) flags |= CF_USE_ROW_TTL;
    out.push_back(flags);
    if (!use_row_ts) h.w_ts(out, c.ts);
    if ((is_deleted || is_expiring) && !use_row_ttl) h.w_ldt(out, ldt_to_long(c.ldt));
    if (is_expiring && !use_row_ttl) h.w_ttl(out, c.ttl);
    if (complex) {  // CellPath: vint length + bytes (CollectionType.java:401-404)
        put_unsigned_vint(out, c.path.size());
        out.insert(out.end(), c.path.begin(), c.path.end());
    }
    if (has_value) put_typed_value(out, complex ? CqlType::BYTES : type, c.value);
}
static Cell read_cell(Reader& r, CqlType type, const LivenessInfo& row_live, const Header& h,
                      bool complex = false) {
    uint8_t flags = r.u8();
    bool has_value = !(flags & CF_HAS_EMPTY_VALUE);
    bool is_deleted = flags & CF_IS_DELETED;
    bool is_expiring = flags & CF_IS_EXPIRING;
    bool use_row_ts = flags & CF_USE_ROW_TIMESTAMP;
    bool use_row_ttl = flags & CF_USE_ROW_TTL;
    Cell c;
    c.ts = use_row_ts ? row_live.ts : h.r_ts(r);
    int64_t ldt = use_row_ttl ? row_live.let
                              : ((is_deleted || is_expiring) ? h.r_ldt(r) : NO_DELETION_TIME);
    c.ttl = use_row_ttl ? row_live.ttl : (is_expiring ? h.r_ttl(r) : NO_TTL);
    c.ldt = ldt_to_u32(ldt);
    if (complex) {
        uint64_t plen = read_unsigned_vint(r);
        c.path = r.take((size_t)plen);
    }
    if (has_value) c.value = read_typed_value(r, complex ? CqlType::BYTES : type);
    return c;
}

// ---- columns subset (Columns.java:503-608) ----
static void put_column_subset(bytes& out, const Row& row, size_t superset_count) {
    // encodeBitmap: bit i set == superset column i MISSING (Columns.java:586-608)
    if (superset_count >= 64) throw std::runtime_error("large column subsets unsupported");
    uint64_t bitmap = 0;
    for (size_t i = 0; i < superset_count; i++)
        if (!row.col_present(i)) bitmap |= 1ULL << i;
    put_unsigned_vint(out, bitmap);
}
static uint64_t read_column_subset_bitmap(Reader& r, size_t superset_count) {
    if (superset_count >= 64) throw std::runtime_error("large column subsets unsupported");
    return read_unsigned_vint(r);  // bit i set == column i missing
}

// ---- row body ----
static void put_row_body(bytes& out, const Row& row, const Header& h, uint8_t flags) {
    if (flags & F_HAS_TIMESTAMP) h.w_ts(out, row.live.ts);
    if (flags & F_HAS_TTL) { h.w_ttl(out, row.live.ttl); h.w_ldt(out, row.live.let); }
    if (flags & F_HAS_DELETION) h.w_dt(out, row.del);
    const auto& cols = row.static_flag ? h.static_cols : h.regular_cols;
    if (!(flags & F_HAS_ALL_COLUMNS)) put_column_subset(out, row, cols.size());
    for (size_t i = 0; i < cols.size(); i++) {
        if (is_complex_type(cols[i].second)) {
            // UnfilteredSerializer.writeComplexColumn (UnfilteredSerializer.java:271-280)
            if (i >= row.complex.size() || !row.complex[i]) continue;
            const ComplexData& cd = *row.complex[i];
            if (flags & F_HAS_COMPLEX_DELETION) h.w_dt(out, cd.del);
            put_unsigned_vint(out, cd.cells.size());
            for (const Cell& c : cd.cells) put_cell(out, c, cols[i].second, row.live, h, true);
        } else if (row.cells[i]) {
            put_cell(out, *row.cells[i], cols[i].second, row.live, h);
        }
    }
}

// row flag computation (UnfilteredSerializer.java:151-185)
static uint8_t row_flags(const Row& row, const Header& h) {
    uint8_t flags = 0;
    size_t present = 0;
    size_t ncols = (row.static_flag ? h.static_cols : h.regular_cols).size();
    for (size_t i = 0; i < ncols; i++) if (row.col_present(i)) present++;
    if (!row.live.empty()) flags |= F_HAS_TIMESTAMP;
    if (row.live.expiring()) flags |= F_HAS_TTL;
    if (!row.del.live()) flags |= F_HAS_DELETION;
    if (present == ncols) flags |= F_HAS_ALL_COLUMNS;
    for (auto& cd : row.complex)
        if (cd && !cd->del.live()) { flags |= F_HAS_COMPLEX_DELETION; break; }
    return flags;
}

void serialize_unfiltered(bytes& out, const Unfiltered& u, const Header& h,
                          uint64_t prev_size) {
    if (u.kind == Unfiltered::MARKER) {
        const Marker& m = u.marker;
        out.push_back(F_IS_MARKER);
        out.push_back((uint8_t)m.kind);            // ClusteringBoundOrBoundary kind ordinal
        put_be16(out, (uint16_t)m.values.size());  // bound size (u16)
        put_clustering_values(out, m.values, h.clustering_types);
        bytes body;
        if (m.boundary()) { h.w_dt(body, m.end_dt); h.w_dt(body, m.start_dt); }
        else h.w_dt(body, m.end_dt);
        // serializedMarkerBodySize INCLUDES sizeof(prev vint) (UnfilteredSerializer.java:394-411)
        put_unsigned_vint(out, body.size() + unsigned_vint_size(prev_size));
        put_unsigned_vint(out, prev_size);
        out.insert(out.end(), body.begin(), body.end());
        return;
    }
    const Row& row = u.row;
    uint8_t flags = row_flags(row, h);
    out.push_back(flags);
    put_clustering_values(out, row.clustering, h.clustering_types);
    bytes body;
    put_row_body(body, row, h, flags);
    // row size = body + sizeof(prev vint) (UnfilteredSerializer.java:193-204)
    put_unsigned_vint(out, body.size() + unsigned_vint_size(prev_size));
    put_unsigned_vint(out, prev_size);
    out.insert(out.end(), body.begin(), body.end());
}

// marker body size check note: serializedMarkerBodySize (UnfilteredSerializer.java:~350)
// = deletion-time sizes + sizeofUnsignedVInt(prev); mirrored above? The reference writes
//   writeUnsignedVInt(serializedMarkerBodySize(..., previousUnfilteredSize, ...))
//   where that size INCLUDES sizeof(prev vint). Verified against fixture in tests.

static Row read_row(Reader& r, const Header& h, uint8_t flags, bool is_static_row) {
    Row row;
    row.static_flag = is_static_row;
    if (!is_static_row)
        row.clustering = read_clustering_values(r, h.clustering_types.size(), h.clustering_types);
    read_unsigned_vint(r);  // row size (skip-aid)
    read_unsigned_vint(r);  // prev size
    if (flags & F_HAS_TIMESTAMP) row.live.ts = h.r_ts(r);
    if (flags & F_HAS_TTL) { row.live.ttl = h.r_ttl(r); row.live.let = h.r_ldt(r); }
    if (flags & F_HAS_DELETION) row.del = h.r_dt(r);
    const auto& cols = is_static_row ? h.static_cols : h.regular_cols;
    uint64_t missing = 0;
    if (!(flags & F_HAS_ALL_COLUMNS)) missing = read_column_subset_bitmap(r, cols.size());
    row.cells.resize(cols.size());
    bool any_cpx = false;
    for (auto& cp : cols) any_cpx |= is_complex_type(cp.second);
    if (any_cpx) row.complex.resize(cols.size());
    for (size_t i = 0; i < cols.size(); i++) {
        if (missing & (1ULL << i)) continue;
        if (is_complex_type(cols[i].second)) {
            ComplexData cd;
            if (flags & F_HAS_COMPLEX_DELETION) cd.del = h.r_dt(r);
            uint64_t n = read_unsigned_vint(r);
            cd.cells.reserve(n);
            for (uint64_t ci = 0; ci < n; ci++)
                cd.cells.push_back(read_cell(r, cols[i].second, row.live, h, true));
            row.complex[i] = std::move(cd);
        } else {
            row.cells[i] = read_cell(r, cols[i].second, row.live, h);
        }
    }
    return row;
}

// ---------------------------------------------------------------------------
// partition serialization (SortedTablePartitionWriter.java:97-166 +
// BigFormatPartitionWriter.java:128-245 + RowIndexEntry.java:460-483,625-647)
// ---------------------------------------------------------------------------
struct IndexInfoC {
    Clustering first, last;
    BoundKind first_kind = CLUSTERING_K, last_kind = CLUSTERING_K;
    uint64_t offset = 0, width = 0;
    DeletionTime end_open_marker;
    bool has_open = false;
};
static void put_clustering_prefix(bytes& out, BoundKind kind, const Clustering& c,
                                  const std::vector<CqlType>& types) {
    // ClusteringPrefix.Serializer.serialize (ClusteringPrefix.java:~420)
    out.push_back((uint8_t)kind);
    if (kind == CLUSTERING_K) put_clustering_values(out, c, types);
    else { put_be16(out, (uint16_t)c.size()); put_clustering_values(out, c, types); }
}
static void put_index_info(bytes& out, const IndexInfoC& ii, const Header& h) {
    // IndexInfo.Serializer (IndexInfo.java:90-118); WIDTH_BASE = 64 KiB
    put_clustering_prefix(out, ii.first_kind, ii.first, h.clustering_types);
    put_clustering_prefix(out, ii.last_kind, ii.last, h.clustering_types);
    put_unsigned_vint(out, ii.offset);
    put_vint(out, (int64_t)ii.width - 65536);
    out.push_back(ii.has_open ? 1 : 0);
    if (ii.has_open) put_deletion_time(out, ii.end_open_marker);
}

void serialize_partition_ex(const Partition& p, const Header& h, bytes& out, bytes& index_out,
                            uint32_t column_index_size, std::vector<IndexInfoC>* out_blocks);
void serialize_partition(const Partition& p, const Header& h, bytes& out, bytes& index_out, uint32_t column_index_size) {
    serialize_partition_ex(p, h, out, index_out, column_index_size, nullptr);
}
void serialize_partition_ex(const Partition& p, const Header& h, bytes& out, bytes& index_out,
                            uint32_t column_index_size, std::vector<IndexInfoC>* out_blocks) {
    size_t initial = out.size();
    put_short_len_bytes(out, p.key);
    put_deletion_time(out, p.del);
    uint64_t header_len = out.size() - initial;
    if (h.has_static()) {
        Row sr = p.static_row;
        sr.static_flag = true;
        sr.cells.resize(h.static_cols.size());
        uint8_t flags = row_flags(sr, h) | F_EXTENSION;
        out.push_back(flags);
        out.push_back(XF_IS_STATIC);
        bytes body;
        put_row_body(body, sr, h, flags);
        put_unsigned_vint(out, body.size() + unsigned_vint_size(0));
        put_unsigned_vint(out, 0);  // previousUnfilteredSize of the static row
        out.insert(out.end(), body.begin(), body.end());
        header_len = out.size() - initial;
    }

    std::vector<IndexInfoC> blocks;
    bool block_open = false;
    IndexInfoC cur;
    DeletionTime open_marker = DT_LIVE;
    uint64_t prev_start = 0;
    uint64_t written = 0;
    for (const Unfiltered& u : p.items) {
        uint64_t pos = out.size() - initial;
        if (!block_open) {
            cur = IndexInfoC{};
            cur.first = u.clustering();
            cur.first_kind = u.kind == Unfiltered::MARKER ? u.marker.kind : CLUSTERING_K;
            cur.offset = pos;
            cur.has_open = !open_marker.live();
            cur.end_open_marker = open_marker;  // will be overwritten at close
            block_open = true;
        }
        serialize_unfiltered(out, u, h, pos - prev_start);
        prev_start = pos;
        written++;
        cur.last = u.clustering();
        cur.last_kind = u.kind == Unfiltered::MARKER ? u.marker.kind : CLUSTERING_K;
        if (u.kind == Unfiltered::MARKER)
            open_marker = u.marker.open(false) ? u.marker.open_dt() : DT_LIVE;
        uint64_t cur_pos = out.size() - initial;
        if (cur_pos - cur.offset >= (uint64_t)column_index_size) {
            cur.width = cur_pos - cur.offset;
            cur.has_open = !open_marker.live();
            cur.end_open_marker = open_marker;
            blocks.push_back(cur);
            block_open = false;
        }
    }
    out.push_back(F_END_OF_PARTITION);
    if (written > 0 && block_open) {
        // BigFormatPartitionWriter.finish() calls addIndexBlock AFTER the
        // end-of-partition byte is written, so the final block's width
        // includes it (currentPosition() - startPosition at that point).
        uint64_t end_position = out.size() - initial;
        cur.width = end_position - cur.offset;
        cur.has_open = !open_marker.live();
        cur.end_open_marker = open_marker;
        blocks.push_back(cur);
    }
    // NOTE on has_open of a CLOSED block: IndexInfo.endOpenMarker is the open marker
    // state at the END of the block (BigFormatPartitionWriter.addIndexBlock:128-134).
    if (out_blocks) *out_blocks = blocks;

    // Index.db entry (BigTableWriter.IndexWriter.append:266-279)
    put_short_len_bytes(index_out, p.key);
    uint64_t position = initial;  // caller must pass a stream where partition starts at `initial`
    if (blocks.size() > 1) {
        // IndexedEntry.serialize (RowIndexEntry.java:625-647)
        bytes infos;
        std::vector<uint32_t> offsets;
        for (auto& b : blocks) {
            offsets.push_back((uint32_t)infos.size());
            put_index_info(infos, b, h);
        }
        bytes fields;
        put_unsigned_vint(fields, header_len);
        put_deletion_time(fields, p.del);
        put_unsigned_vint(fields, blocks.size());
        uint64_t size = fields.size() + infos.size() + offsets.size() * 4;
        put_unsigned_vint(index_out, position);
        put_unsigned_vint(index_out, size);
        index_out.insert(index_out.end(), fields.begin(), fields.end());
        index_out.insert(index_out.end(), infos.begin(), infos.end());
        for (uint32_t o : offsets) put_be32(index_out, o);
    } else {
        put_unsigned_vint(index_out, position);
        put_unsigned_vint(index_out, 0);  // RowIndexEntry.serialize (:468-473)
    }
}

// ---------------------------------------------------------------------------
// chunked compression (CompressedSequentialWriter.flushData:140-206,
// LZ4Compressor.java:118-134, ChecksumWriter.java:62-104)
// ---------------------------------------------------------------------------
ChunkedOut chunk_compress(const bytes& raw, const CompressionParams& cp) {
    if (cp.algo != Compressor::LZ4 && cp.algo != Compressor::SNAPPY)
        throw std::runtime_error("unsupported compressor");
    ChunkedOut co;
    size_t nchunks = (raw.size() + cp.chunk_len - 1) / cp.chunk_len;
    bytes tmp(4 + std::max<size_t>(LZ4_compressBound(cp.chunk_len),
                                   snappy_ref_max_compressed_length(cp.chunk_len)));
    for (size_t i = 0; i < nchunks; i++) {
        size_t off = i * cp.chunk_len;
        size_t len = std::min((size_t)cp.chunk_len, raw.size() - off);
        uint32_t total;
        if (cp.algo == Compressor::SNAPPY) {
            // SnappyCompressor.compress (SnappyCompressor.java:82-86): the
            // chunk payload is one raw snappy block (snappy carries its own
            // leading uncompressed-length varint; no extra header)
            size_t csz = tmp.size();
            if (!snappy_ref_compress((const char*)raw.data() + off, len, (char*)tmp.data(), &csz))
                throw std::runtime_error("snappy compress failed");
            total = (uint32_t)csz;
        } else {
        // 4-byte LITTLE-endian uncompressed length + raw LZ4 block
        tmp[0] = (uint8_t)len; tmp[1] = (uint8_t)(len >> 8);
        tmp[2] = (uint8_t)(len >> 16); tmp[3] = (uint8_t)(len >> 24);
        int csz = LZ4_compress_default((const char*)raw.data() + off, (char*)tmp.data() + 4,
                                       (int)len, (int)tmp.size() - 4);
        if (csz <= 0) throw std::runtime_error("LZ4_compress_default failed");
        total = (uint32_t)csz + 4;
        }
        if (total >= cp.max_compressed_len)
            throw std::runtime_error("store-uncompressed fallback unsupported (default params never hit it)");
        co.offsets.push_back(co.file.size());
        co.file.insert(co.file.end(), tmp.begin(), tmp.begin() + total);
        uint32_t crc = crc32(tmp.data(), total);  // CRC of the COMPRESSED bytes
        put_be32(co.file, crc);
    }
    return co;
}

bytes make_compression_info(const CompressionParams& cp, uint64_t data_len,
                            const std::vector<uint64_t>& offsets) {
    // CompressionMetadata.Writer.writeHeader + doPrepare (CompressionMetadata.java:375-440)
    bytes out;
    put_utf(out, cp.algo == Compressor::LZ4 ? "LZ4Compressor" : "SnappyCompressor");
    put_be32(out, 0);  // option count (LZ4 default: none — LZ4Compressor.compressionOptions empty)
    put_be32(out, cp.chunk_len);
    put_be32(out, cp.max_compressed_len);
    put_be64(out, data_len);
    put_be32(out, (uint32_t)offsets.size());
    for (uint64_t o : offsets) put_be64(out, o);
    return out;
}

// ---------------------------------------------------------------------------
// bloom filter (utils/BloomFilter.java:79-122, BloomCalculations.java,
// FilterFactory.java, OffHeapBitSet.java:87-120; fp chance fixed at 0.01)
// ---------------------------------------------------------------------------
struct BloomSpec { int k; int buckets; };
static BloomSpec compute_bloom_spec(int max_buckets, double max_fp) {
    static const std::vector<std::vector<double>> probs = {
        {1.0}, {1.0, 1.0},
        {1.0, 0.393, 0.400},
        {1.0, 0.283, 0.237, 0.253},
        {1.0, 0.221, 0.155, 0.147, 0.160},
        {1.0, 0.181, 0.109, 0.092, 0.092, 0.101},
        {1.0, 0.154, 0.0804, 0.0609, 0.0561, 0.0578, 0.0638},
        {1.0, 0.133, 0.0618, 0.0423, 0.0359, 0.0347, 0.0364},
        {1.0, 0.118, 0.0489, 0.0306, 0.024, 0.0217, 0.0216, 0.0229},
        {1.0, 0.105, 0.0397, 0.0228, 0.0166, 0.0141, 0.0133, 0.0135, 0.0145},
        {1.0, 0.0952, 0.0329, 0.0174, 0.0118, 0.00943, 0.00844, 0.00819, 0.00846},
        {1.0, 0.0869, 0.0276, 0.0136, 0.00864, 0.0065, 0.00552, 0.00513, 0.00509},
        {1.0, 0.08, 0.0236, 0.0108, 0.00646, 0.00459, 0.00371, 0.00329, 0.00314},
        {1.0, 0.074, 0.0203, 0.00875, 0.00492, 0.00332, 0.00255, 0.00217, 0.00199, 0.00194},
        {1.0, 0.0689, 0.0177, 0.00718, 0.00381, 0.00244, 0.00179, 0.00146, 0.00129, 0.00121, 0.0012},
        {1.0, 0.0645, 0.0156, 0.00596, 0.003, 0.00183, 0.00128, 0.001, 0.000852, 0.000775, 0.000744},
        {1.0, 0.0606, 0.0138, 0.005, 0.00239, 0.00139, 0.000935, 0.000702, 0.000574, 0.000505, 0.00047, 0.000459},
        {1.0, 0.0571, 0.0123, 0.00423, 0.00193, 0.00107, 0.000692, 0.000499, 0.000394, 0.000335, 0.000302, 0.000287, 0.000284},
        {1.0, 0.054, 0.0111, 0.00362, 0.00158, 0.000839, 0.000519, 0.00036, 0.000275, 0.000226, 0.000198, 0.000183, 0.000176},
        {1.0, 0.0513, 0.00998, 0.00312, 0.0013, 0.000663, 0.000394, 0.000264, 0.000194, 0.000155, 0.000132, 0.000118, 0.000111, 0.000109},
        {1.0, 0.0488, 0.00906, 0.0027, 0.00108, 0.00053, 0.000303, 0.000196, 0.00014, 0.000108, 8.89e-05, 7.77e-05, 7.12e-05, 6.79e-05, 6.71e-05},
    };
    auto optK = [&](int b) {
        double mn = 1e300; int kk = 1;
        for (size_t j = 0; j < probs[b].size(); j++)
            if (probs[b][j] < mn) { mn = probs[b][j]; kk = std::max(1, (int)j); }
        return kk;
    };
    if (max_fp >= probs[2][1]) return {optK(2), 2};
    int b = 2, k = optK(2);
    while (probs[b][k] > max_fp) { b++; k = optK(b); }
    while (probs[b][k - 1] <= max_fp) k--;
    (void)max_buckets;
    return {k, b};
}

struct Bloom {
    int hash_count;
    bytes bits;  // capacity = words*64 bits; bit i -> bits[i>>3] & 1<<(i&7)
    void add(const bytes& key) {
        uint64_t h[2];
        murmur3_128_cassandra(key.data(), key.size(), 0, h);
        uint64_t max = (uint64_t)bits.size() * 8;
        int64_t base = (int64_t)h[1], inc = (int64_t)h[0];
        for (int i = 0; i < hash_count; i++) {
            int64_t m = base % (int64_t)max;
            uint64_t idx = (uint64_t)((m ^ (m >> 63)) - (m >> 63));  // FBUtilities.abs
            bits[idx >> 3] |= 1u << (idx & 7);
            base += inc;
        }
    }
};
static Bloom make_bloom(uint64_t num_keys, double fp /*=0.01*/) {
    int maxb = 20;
    BloomSpec spec = compute_bloom_spec(maxb, fp);
    uint64_t num_bits = num_keys * (uint64_t)spec.buckets + 20;  // BITSET_EXCESS
    uint64_t words = ((num_bits - 1) >> 6) + 1;
    Bloom b;
    b.hash_count = spec.k;
    b.bits.assign(words * 8, 0);
    return b;
}
static bytes serialize_bloom(const Bloom& b) {
    bytes out;
    put_be32(out, (uint32_t)b.hash_count);
    put_be32(out, (uint32_t)(b.bits.size() / 8));
    out.insert(out.end(), b.bits.begin(), b.bits.end());
    return out;
}

// ---------------------------------------------------------------------------
// Statistics.db (MetadataSerializer.java:52-115 + component serializers)
// ---------------------------------------------------------------------------
static void put_type_str(bytes& out, CqlType t) {
    std::string s = cql_type_name(t);
    put_unsigned_vint(out, s.size());
    out.insert(out.end(), s.begin(), s.end());
}
static bytes serialize_header_component(const Header& h) {
    // SerializationHeader.Component serializer (SerializationHeader.java:~380)
    bytes out;
    put_unsigned_vint(out, (uint64_t)(h.stats.min_ts - TIMESTAMP_EPOCH));
    put_unsigned_vint(out, (uint64_t)(int64_t)(int32_t)(h.stats.min_ldt - DELETION_TIME_EPOCH));
    put_unsigned_vint(out, (uint64_t)(int64_t)(int32_t)(h.stats.min_ttl - 0));
    put_type_str(out, h.key_type);
    put_unsigned_vint(out, h.clustering_types.size());
    for (CqlType t : h.clustering_types) put_type_str(out, t);
    for (auto cols : {&h.static_cols, &h.regular_cols}) {
        put_unsigned_vint(out, cols->size());
        for (auto& [name, t] : *cols) { put_vint_len_bytes(out, name); put_type_str(out, t); }
    }
    return out;
}

// EstimatedHistogram with the standard bucket offsets (EstimatedHistogram.java
// newOffsets: 1, 2, ... growing by 1.2x) — used for partition-size & column-count.
struct EstHist {
    std::vector<int64_t> offsets;  // EstimatedHistogram.newOffsets (EstimatedHistogram.java:91-107)
    std::vector<int64_t> buckets;
    explicit EstHist(int size) {
        int64_t last = 1;
        offsets.push_back(1);
        for (int i = 1; i < size; i++) {
            int64_t next = (int64_t)llround((double)last * 1.2);
            if (next == last) next++;
            offsets.push_back(next);
            last = next;
        }
        buckets.assign(offsets.size() + 1, 0);
    }
    void add(uint64_t n) {
        auto it = std::lower_bound(offsets.begin(), offsets.end(), (int64_t)n);
        buckets[it - offsets.begin()]++;
    }
};
static void put_est_hist(bytes& out, const EstHist& h) {
    // EstimatedHistogram.serializer (EstimatedHistogram.java): int count then
    // per bucket (long offset[i==0?0:i-1], long bucket[i])
    put_be32(out, (uint32_t)h.buckets.size());
    for (size_t i = 0; i < h.buckets.size(); i++) {
        put_be64(out, (uint64_t)h.offsets[i == 0 ? 0 : i - 1]);
        put_be64(out, (uint64_t)h.buckets[i]);
    }
}

struct StatsComponentInput {
    // sizes per MetadataCollector.defaultPartitionSizeHistogram (155) and
    // defaultCellPerPartitionCountHistogram (118), MetadataCollector.java:60-69
    EstHist partition_size{155};
    EstHist cells_per_partition{118};
    StatsMins mins;
    double compression_ratio = -1;
    bytes first_key, last_key;
    std::vector<uint64_t> key_hashes;  // hash2_64 per partition key (HLL)
    std::map<uint32_t, uint32_t> tombstone_hist;  // ldt(seconds)->count, ≤100 bins
    double token_space_coverage = 0;
    size_t clustering_count = 0;
    CqlType clustering_type = CqlType::LONG;
    bool has_partition_deletions = false;
};
static bytes serialize_stats_component(const StatsComponentInput& s) {
    // StatsMetadata.serializer.serialize for version `oa` (StatsMetadata.java:402-512)
    bytes out;
    put_est_hist(out, s.partition_size);
    put_est_hist(out, s.cells_per_partition);
    put_be64(out, (uint64_t)-1LL); put_be32(out, 0);  // commitLogUpperBound = NONE(-1,0)
    put_be64(out, (uint64_t)s.mins.min_timestamp);
    put_be64(out, (uint64_t)s.mins.max_timestamp);
    put_be32(out, ldt_to_u32(s.mins.min_ldt));   // hasUIntDeletionTime
    put_be32(out, ldt_to_u32(s.mins.max_ldt));
    put_be32(out, (uint32_t)s.mins.min_ttl);
    put_be32(out, (uint32_t)s.mins.max_ttl);
    uint64_t cr; double crv = s.compression_ratio; memcpy(&cr, &crv, 8); put_be64(out, cr);
    // TombstoneHistogram (streamhist/TombstoneHistogram.java:77-95): maxBinSize=100
    put_be32(out, 100);
    put_be32(out, (uint32_t)s.tombstone_hist.size());
    for (auto& [pt, cnt] : s.tombstone_hist) { put_be64(out, pt); put_be32(out, cnt); }
    put_be32(out, 0);               // sstableLevel
    put_be64(out, 0);               // repairedAt = UNREPAIRED_SSTABLE
    // improvedMinMax: typeSerializer.serializeList + Slice (spec'd with the GPU
    // writer: type strings then BOTTOM..TOP bounds without values)
    put_unsigned_vint(out, s.clustering_count);
    for (size_t i = 0; i < s.clustering_count; i++) put_type_str(out, s.clustering_type);
    out.push_back((uint8_t)INCL_START); put_be16(out, 0);  // start bound, 0 values
    out.push_back((uint8_t)INCL_END);   put_be16(out, 0);  // end bound, 0 values
    out.push_back(0);                                       // hasLegacyCounterShards=false
    put_be64(out, (uint64_t)s.mins.total_columns_set);
    put_be64(out, (uint64_t)s.mins.total_rows);
    put_be64(out, (uint64_t)-1LL); put_be32(out, 0);  // commitLogLowerBound = NONE
    put_be32(out, 0);                                  // commitLogIntervals: empty set
    out.push_back(0);                                  // pendingRepair = null
    out.push_back(0);                                  // isTransient = false
    out.push_back(0);                                  // originatingHostId = null
    out.push_back(s.has_partition_deletions ? 1 : 0);  // hasPartitionLevelDeletionsPresenceMarker
    put_vint_len_bytes(out, s.first_key);
    put_vint_len_bytes(out, s.last_key);
    uint64_t ts; double tsc = s.token_space_coverage; memcpy(&ts, &tsc, 8); put_be64(out, ts);
    return out;
}

// HyperLogLogPlus of the partition-key hash2_64 values (clearspring
// stream-lib 2.5.x, p=13 sp=25 — MetadataCollector's `new HyperLogLogPlus
// (13, 25)`), COMPACTION component. SPARSE (format 1) while the distinct
// sparse-index set stays under the 0.75*m conversion threshold and no key
// needs the flagged rho encoding (a 2^-(sp-p) event): entries are
// (hash >>> (64-sp)) << 1, serialized as delta varints over the sorted set
// — byte-pinned against the reference's own oa fixture. Otherwise NORMAL
// (format 0): 2^p five-bit registers, six per 32-bit word
// (RegisterSet.REGISTER_SIZE=5, LOG2_BITS_PER_WORD=6), register =
// max(nlz((h << p) | 1 << (p-1)) + 1), words big-endian.
static bytes serialize_compaction_component(const std::vector<uint64_t>& key_hashes) {
    constexpr int P = 13, SP = 25;
    bytes hll;
    put_be32(hll, (uint32_t)-2);  // -VERSION
    auto put_varint = [&](uint32_t v) { while (v >= 0x80) { hll.push_back((uint8_t)(v | 0x80)); v >>= 7; } hll.push_back((uint8_t)v); };
    put_varint(P); put_varint(SP);
    std::set<uint32_t> sparse;
    bool flagged = false;
    const uint32_t threshold = (uint32_t)((1u << P) * 3 / 4);
    for (uint64_t h : key_hashes) {
        uint32_t sidx = (uint32_t)(h >> (64 - SP));
        if ((sidx & ((1u << (SP - P)) - 1)) == 0) { flagged = true; break; }
        sparse.insert(sidx << 1);
        if (sparse.size() > threshold) break;
    }
    if (!flagged && sparse.size() <= threshold && key_hashes.size() <= threshold) {
        put_varint(1 /*SPARSE*/);
        put_varint((uint32_t)sparse.size());
        uint32_t prev = 0;
        for (uint32_t v : sparse) { put_varint(v - prev); prev = v; }
    } else {
        put_varint(0 /*NORMAL*/);
        std::vector<uint8_t> regs(1u << P, 0);
        for (uint64_t h : key_hashes) {
            uint32_t idx = (uint32_t)(h >> (64 - P));
            uint64_t w = (h << P) | (1ull << (P - 1));
            uint8_t rho = (uint8_t)(__builtin_clzll(w) + 1);
            if (rho > regs[idx]) regs[idx] = rho;
        }
        uint32_t bits = (1u << P) / 6;
        uint32_t reg_ints = (bits % 32 == 0) ? bits : bits + 1;  // RegisterSet.getSizeForCount
        std::vector<uint32_t> M(reg_ints, 0);
        for (uint32_t i = 0; i < (1u << P); i++)
            M[i / 6] |= (uint32_t)regs[i] << (5 * (i % 6));
        put_varint(reg_ints * 4);
        for (uint32_t wv : M) put_be32(hll, wv);
    }
    bytes out;
    put_be32(out, (uint32_t)hll.size());
    out.insert(out.end(), hll.begin(), hll.end());
    return out;
}

static bytes serialize_statistics(const Header& h, const StatsComponentInput& s) {
    // MetadataSerializer.serialize (na+ checksummed; MetadataSerializer.java:67-115)
    // component order by MetadataType ordinal: VALIDATION=0 COMPACTION=1 STATS=2 HEADER=3
    bytes validation;
    put_utf(validation, "org.apache.cassandra.dht.Murmur3Partitioner");
    uint64_t fp; double fpv = 0.01; memcpy(&fp, &fpv, 8); put_be64(validation, fp);
    bytes compaction = serialize_compaction_component(s.key_hashes);
    bytes stats = serialize_stats_component(s);
    bytes header = serialize_header_component(h);
    const bytes* comps[4] = {&validation, &compaction, &stats, &header};

    bytes out;
    uint32_t crc = 0;
    put_be32(out, 4);
    crc = crc32_update_int(0, 4);
    put_be32(out, crc);
    uint32_t pos = 4 + 8 * 4 + 2 * 4;  // count + toc + 2 crcs
    uint32_t toc_crc = crc;            // CRC continues over toc after count
    for (int i = 0; i < 4; i++) {
        put_be32(out, i);
        toc_crc = crc32_update_int(toc_crc, i);
        put_be32(out, pos);
        toc_crc = crc32_update_int(toc_crc, pos);
        pos += comps[i]->size() + 4;
    }
    put_be32(out, toc_crc);
    for (int i = 0; i < 4; i++) {
        out.insert(out.end(), comps[i]->begin(), comps[i]->end());
        put_be32(out, crc32(comps[i]->data(), comps[i]->size()));
    }
    return out;
}

// ---------------------------------------------------------------------------
// whole-sstable writer
// ---------------------------------------------------------------------------
// ---------------------------------------------------------------------------
// BTI (`da`) index assembly: Partitions.db + Rows.db from the SAME
// column_index_size blocks as the big promoted index (rowIndexBlockSize ==
// column_index_size, BtiFormatPartitionWriter.java:52). Row-index entry b
// carries (block b's start offset within the partition, the deletion open at
// its start == block b-1's end-open); a partition gets a row index only when
// it has >1 blocks (BtiFormatPartitionWriter.finish:100-109); the final
// entry is nudge(prevMax, |common(prevMax, prevSep)|) with payload
// (partition length - 1 == the END_OF_PARTITION byte's offset, LIVE) —
// pinned against the legacy_da fixtures (cmd_roundtrip Stats/Partitions/
// Rows comparisons).
static void build_bti_index(const SSTable& t, const std::vector<uint64_t>& part_pos,
                            const std::vector<std::vector<IndexInfoC>>& part_blocks,
                            uint64_t data_len, WriterOut& w) {
    bytes rows;
    std::vector<BtiKeyEntry> pes;
    auto enc = [&](BoundKind k, const Clustering& c) {
        return bti_byte_comparable_clustering(c, t.header.clustering_types, k);
    };
    auto payload = [](uint64_t off, bool has_open, const DeletionTime& od, int* pb, bytes* pay) {
        // SizedInts.nonZeroSize: SIGNED size — the leading bit stays clear
        int ob = 1;
        while (off >> (8 * ob - 1)) ob++;
        *pb = ob | (has_open ? 8 : 0);
        for (int b = ob - 1; b >= 0; b--) pay->push_back((uint8_t)(off >> (8 * b)));
        if (has_open) {
            if (od.live()) {
                pay->push_back(0x80);
            } else {
                for (int b = 7; b >= 0; b--) pay->push_back((uint8_t)((uint64_t)od.mfda >> (8 * b)));
                for (int b = 3; b >= 0; b--) pay->push_back((uint8_t)(od.ldt >> (8 * b)));
            }
        }
    };
    for (size_t i = 0; i < t.parts.size(); i++) {
        const Partition& p = t.parts[i];
        uint64_t part_end = (i + 1 < t.parts.size() ? part_pos[i + 1] : data_len) - part_pos[i];
        const auto& blocks = part_blocks[i];
        int64_t idxpos;
        if (blocks.size() > 1) {
            BtiRowIndexBlockSpec spec;
            spec.partition_key = p.key;
            spec.data_pos = part_pos[i];
            spec.block_count = blocks.size();
            spec.partition_del = p.del;
            bytes prev_sep, prev_max;
            for (size_t b = 0; b < blocks.size(); b++) {
                bytes key;
                if (b > 0)
                    key = bti_separator_gt(enc(blocks[b - 1].last_kind, blocks[b - 1].last),
                                           enc(blocks[b].first_kind, blocks[b].first));
                bool has_od = b > 0 && blocks[b - 1].has_open;
                DeletionTime od = has_od ? blocks[b - 1].end_open_marker : DT_LIVE;
                int pb;
                bytes pay;
                payload(blocks[b].offset, has_od, od, &pb, &pay);
                spec.entries.push_back({key, pb, pay});
                prev_sep = key;
                prev_max = enc(blocks[b].last_kind, blocks[b].last);
            }
            size_t cm = 0;
            while (cm < prev_max.size() && cm < prev_sep.size() && prev_max[cm] == prev_sep[cm]) cm++;
            int pb;
            bytes pay;
            payload(part_end - 1, false, DT_LIVE, &pb, &pay);
            spec.entries.push_back({bti_nudge(prev_max, cm), pb, pay});
            idxpos = (int64_t)append_bti_row_index(rows, spec);
        } else {
            idxpos = ~(int64_t)part_pos[i];
        }
        BtiKeyEntry e;
        e.byte_comparable = t.partitioner == Partitioner::MURMUR3
                                ? bti_byte_comparable_m3(p.token, p.key)
                                : bti_byte_comparable_bop(p.key);
        e.raw_key = p.key;
        uint64_t h2[2];
        murmur3_128_cassandra(p.key.data(), p.key.size(), 0, h2);
        e.hash_bits = (uint8_t)h2[1];  // DecoratedKey.filterHashLowerBits (IFilter.java:34-40)
        e.idxpos = idxpos;
        pes.push_back(e);
    }
    w.rows_db = std::move(rows);
    w.partitions_db = write_bti_partitions(pes);
}

WriterOut write_sstable(const SSTable& t, bool bti) {
    WriterOut w;
    w.bti = bti;
    std::vector<uint64_t> part_pos;
    std::vector<std::vector<IndexInfoC>> part_blocks;
    bytes data_raw;
    Bloom bloom = make_bloom(t.parts.size(), 0.01);
    StatsComponentInput st;
    st.mins = StatsMins{};
    uint64_t total_cells = 0, total_rows = 0, total_cols_set = 0;
    bool has_partition_deletions = false;
    std::vector<uint64_t> index_entry_pos;
    for (const Partition& p : t.parts) {
        size_t before = data_raw.size();
        part_pos.push_back(before);
        part_blocks.emplace_back();
        index_entry_pos.push_back(w.index_db.size());
        serialize_partition_ex(p, t.header, data_raw, w.index_db, t.column_index_size,
                               bti ? &part_blocks.back() : nullptr);
        bloom.add(p.key);
        st.partition_size.add(data_raw.size() - before);
        uint64_t cells = 0;
        if (!p.del.live()) has_partition_deletions = true;
        for (auto& u : p.items) {
            if (u.kind != Unfiltered::ROW) continue;
            total_rows++;
            // Rows.collectStats: totalColumnsSet counts COLUMNS (a complex
            // column once when it has >=1 cell); the per-partition histogram
            // counts CELLS (each complex cell) — Rows.java:60-82
            for (auto& c : u.row.cells) if (c) { cells++; total_cols_set++; }
            for (auto& cd : u.row.complex)
                if (cd && !cd->cells.empty()) { cells += cd->cells.size(); total_cols_set++; }
        }
        total_cells += cells;
        st.cells_per_partition.add(cells);
    }
    // mins/maxes per MetadataCollector semantics (update per liveness/cell/deletion)
    auto upd_ts = [&](int64_t ts) { if (ts == NO_TIMESTAMP) return; st.mins.min_timestamp = std::min(st.mins.min_timestamp, ts); st.mins.max_timestamp = std::max(st.mins.max_timestamp, ts); };
    auto upd_ldt = [&](int64_t l) { st.mins.min_ldt = std::min(st.mins.min_ldt, l); st.mins.max_ldt = std::max(st.mins.max_ldt, l); };
    auto upd_ttl = [&](int32_t ttl) { st.mins.min_ttl = std::min(st.mins.min_ttl, ttl); st.mins.max_ttl = std::max(st.mins.max_ttl, ttl); };
    auto upd_tomb = [&](uint32_t ldt_sec) { st.tombstone_hist[ldt_sec]++; };
    for (const Partition& p : t.parts) {
        if (!p.del.live()) { upd_ts(p.del.mfda); upd_ldt(ldt_to_long(p.del.ldt)); upd_tomb(p.del.ldt); }
        for (auto& u : p.items) {
            if (u.kind == Unfiltered::MARKER) {
                const Marker& m = u.marker;
                if (m.boundary()) { for (auto* d : {&m.end_dt, &m.start_dt}) { upd_ts(d->mfda); upd_ldt(ldt_to_long(d->ldt)); upd_tomb(d->ldt); } }
                else { upd_ts(m.end_dt.mfda); upd_ldt(ldt_to_long(m.end_dt.ldt)); upd_tomb(m.end_dt.ldt); }
                continue;
            }
            const Row& r = u.row;
            if (!r.live.empty()) { upd_ts(r.live.ts); if (r.live.expiring()) { upd_ttl(r.live.ttl); upd_ldt(r.live.let); } else upd_ldt(NO_DELETION_TIME); }
            if (!r.del.live()) { upd_ts(r.del.mfda); upd_ldt(ldt_to_long(r.del.ldt)); upd_tomb(r.del.ldt); }
            for (auto& c : r.cells) {
                if (!c) continue;
                upd_ts(c->ts);
                if (c->tombstone()) { upd_ldt(ldt_to_long(c->ldt)); upd_tomb(c->ldt); }
                else if (c->expiring()) { upd_ldt(ldt_to_long(c->ldt)); upd_ttl(c->ttl); }
                else upd_ldt(NO_DELETION_TIME);
            }
            for (auto& cd : r.complex) {
                if (!cd) continue;
                // MetadataCollector.update(complexDeletion) — Rows.java:75
                if (!cd->del.live()) { upd_ts(cd->del.mfda); upd_ldt(ldt_to_long(cd->del.ldt)); upd_tomb(cd->del.ldt); }
                for (const Cell& c : cd->cells) {
                    upd_ts(c.ts);
                    if (c.tombstone()) { upd_ldt(ldt_to_long(c.ldt)); upd_tomb(c.ldt); }
                    else if (c.expiring()) { upd_ldt(ldt_to_long(c.ldt)); upd_ttl(c.ttl); }
                    else upd_ldt(NO_DELETION_TIME);
                }
            }
        }
    }
    if (st.mins.min_timestamp == INT64_MAX) { st.mins.min_timestamp = 0; st.mins.max_timestamp = 0; }
    if (st.mins.min_ttl == INT32_MAX) st.mins.min_ttl = 0;
    st.mins.total_rows = total_rows;
    st.mins.total_columns_set = total_cols_set;
    st.clustering_count = t.header.clustering_types.size();
    if (st.clustering_count) st.clustering_type = t.header.clustering_types[0];
    if (!t.parts.empty()) { st.first_key = t.parts.front().key; st.last_key = t.parts.back().key; }
    st.key_hashes.reserve(t.parts.size());
    for (const Partition& p : t.parts)
        st.key_hashes.push_back(murmur2_64_cassandra(p.key.data(), p.key.size(), 0));

    ChunkedOut co = chunk_compress(data_raw, t.comp);
    // compressionRatio = compressedSize/uncompressedSize where compressedSize
    // excludes the per-chunk CRC words (CompressedSequentialWriter.java:158,
    // chunkOffset += compressedLength + 4) — matches the product's writer.
    st.compression_ratio = data_raw.empty() ? -1.0
        : (double)(co.file.size() - 4 * co.offsets.size()) / (double)data_raw.size();
    w.data_db = std::move(co.file);
    w.compression_info = make_compression_info(t.comp, data_raw.size(), co.offsets);
    w.filter = serialize_bloom(bloom);
    w.digest = bytes();
    {
        std::string d = std::to_string(crc32(w.data_db.data(), w.data_db.size()));
        w.digest.assign(d.begin(), d.end());
    }
    st.has_partition_deletions = has_partition_deletions;
    w.statistics = serialize_statistics(t.header, st);
    // Summary.db: IndexSummary at BASE_SAMPLING_LEVEL — entries for keys
    // 0, 128, 256, ... (IndexSummaryBuilder.maybeAddEntry with empty
    // Downsampling start points at full sampling), serialized per
    // IndexSummary.IndexSummarySerializer.serialize: BE header
    // (minIndexInterval, offsetCount, offHeapSize, samplingLevel,
    // sizeAtFullSampling), then the off-heap image in NATIVE (LE) order —
    // per-entry offsets rebased by 4*offsetCount, then key bytes + LE
    // position of the entry in Index.db.
    {
        bytes& s = w.summary;
        const uint32_t MIN_INTERVAL = 128;
        std::vector<uint64_t> sampled;
        for (uint64_t k = 0; k < t.parts.size(); k += MIN_INTERVAL) sampled.push_back(k);
        uint32_t cnt = (uint32_t)sampled.size();
        bytes entries;
        std::vector<uint32_t> offs;
        for (uint64_t k : sampled) {
            offs.push_back((uint32_t)entries.size());
            const bytes& kb = t.parts[k].key;
            entries.insert(entries.end(), kb.begin(), kb.end());
            uint64_t pos = index_entry_pos[k];
            for (int b = 0; b < 8; b++) entries.push_back((uint8_t)(pos >> (8 * b)));
        }
        put_be32(s, MIN_INTERVAL);
        put_be32(s, cnt);
        put_be64(s, 4ull * cnt + entries.size());
        put_be32(s, 128);                        // samplingLevel = BASE
        put_be32(s, cnt);                        // sizeAtFullSampling
        for (uint32_t o : offs) {
            uint32_t v = o + 4 * cnt;
            for (int b = 0; b < 4; b++) s.push_back((uint8_t)(v >> (8 * b)));  // LE
        }
        s.insert(s.end(), entries.begin(), entries.end());
        if (!t.parts.empty()) {
            // first/last key with 4-byte BE lengths (SSTableReader.saveSummary:
            // ByteBufferUtil.writeWithLength)
            put_be32(s, (uint32_t)t.parts.front().key.size());
            s.insert(s.end(), t.parts.front().key.begin(), t.parts.front().key.end());
            put_be32(s, (uint32_t)t.parts.back().key.size());
            s.insert(s.end(), t.parts.back().key.begin(), t.parts.back().key.end());
        }
        std::string toc = w.bti ? "Data.db\nStatistics.db\nDigest.crc32\nTOC.txt\nCompressionInfo.db\nFilter.db\nPartitions.db\nRows.db\n"
                                : "Data.db\nStatistics.db\nDigest.crc32\nTOC.txt\nCompressionInfo.db\nFilter.db\nIndex.db\nSummary.db\n";
        w.toc.assign(toc.begin(), toc.end());
    }
    if (bti) build_bti_index(t, part_pos, part_blocks, data_raw.size(), w);
    w.uncompressed_data_len = data_raw.size();
    w.partition_count = t.parts.size();
    return w;
}

void write_components(const WriterOut& w, const std::string& base) {
    write_file(base + "-Data.db", w.data_db);
    if (w.bti) {
        write_file(base + "-Partitions.db", w.partitions_db);
        write_file(base + "-Rows.db", w.rows_db);
    } else {
        write_file(base + "-Index.db", w.index_db);
        write_file(base + "-Summary.db", w.summary);
    }
    write_file(base + "-CompressionInfo.db", w.compression_info);
    write_file(base + "-Filter.db", w.filter);
    write_file(base + "-Digest.crc32", w.digest);
    write_file(base + "-Statistics.db", w.statistics);
    write_file(base + "-TOC.txt", w.toc);
}

// ---------------------------------------------------------------------------
// reader
// ---------------------------------------------------------------------------
struct CompressionInfo {
    CompressionParams params;
    uint64_t data_len;
    std::vector<uint64_t> offsets;
};
static CompressionInfo read_compression_info(const bytes& b) {
    Reader r(b);
    uint16_t nlen = r.be16();
    bytes name = r.take(nlen);
    std::string algo((char*)name.data(), name.size());
    CompressionInfo ci;
    if (algo == "LZ4Compressor") ci.params.algo = Compressor::LZ4;
    else if (algo == "SnappyCompressor") ci.params.algo = Compressor::SNAPPY;
    else throw std::runtime_error("unsupported compressor " + algo);
    uint32_t opts = r.be32();
    for (uint32_t i = 0; i < opts; i++) { r.take(r.be16()); r.take(r.be16()); }
    ci.params.chunk_len = r.be32();
    ci.params.max_compressed_len = r.be32();
    ci.data_len = r.be64();
    uint32_t n = r.be32();
    for (uint32_t i = 0; i < n; i++) ci.offsets.push_back(r.be64());
    return ci;
}

static bytes decompress_data(const bytes& file, const CompressionInfo& ci) {
    bytes out;
    out.reserve(ci.data_len);
    for (size_t i = 0; i < ci.offsets.size(); i++) {
        uint64_t off = ci.offsets[i];
        uint64_t end = (i + 1 < ci.offsets.size()) ? ci.offsets[i + 1] : file.size();
        if (end < off + 4 + 4) throw std::runtime_error("bad chunk bounds");
        uint64_t comp_len = end - off - 4;  // excludes trailing CRC
        uint32_t crc_stored = 0;
        for (int k = 0; k < 4; k++) crc_stored = (crc_stored << 8) | file[off + comp_len + k];
        uint32_t crc_calc = crc32(file.data() + off, comp_len);
        if (crc_stored != crc_calc) throw std::runtime_error("chunk CRC mismatch");
        uint64_t want = std::min<uint64_t>(ci.params.chunk_len, ci.data_len - out.size());
        if (ci.params.algo == Compressor::LZ4) {
            uint32_t ulen = file[off] | (file[off + 1] << 8) | (file[off + 2] << 16) | ((uint32_t)file[off + 3] << 24);
            if (ulen != want) throw std::runtime_error("chunk length header mismatch");
            size_t prev = out.size();
            out.resize(prev + ulen);
            int got = LZ4_decompress_safe((const char*)file.data() + off + 4, (char*)out.data() + prev,
                                          (int)(comp_len - 4), (int)ulen);
            if (got != (int)ulen) throw std::runtime_error("LZ4 decode failed");
        } else {
            // raw snappy block: decoded length from its own varint header
            size_t ulen = 0;
            if (!snappy_ref_uncompressed_length((const char*)file.data() + off, comp_len, &ulen) ||
                ulen != want)
                throw std::runtime_error("snappy chunk length mismatch");
            size_t prev = out.size();
            out.resize(prev + ulen);
            if (!snappy_ref_uncompress((const char*)file.data() + off, comp_len,
                                       (char*)out.data() + prev))
                throw std::runtime_error("snappy decode failed");
        }
    }
    if (out.size() != ci.data_len) throw std::runtime_error("data length mismatch");
    return out;
}

struct StatisticsFile {
    Header header;
    StatsMins mins;
    Partitioner partitioner = Partitioner::MURMUR3;
};
static StatisticsFile read_statistics(const bytes& b) {
    Reader r(b);
    uint32_t count = r.be32();
    r.be32();  // crc
    std::map<uint32_t, uint32_t> toc;
    for (uint32_t i = 0; i < count; i++) {
        uint32_t type = r.be32();
        uint32_t pos = r.be32();
        toc[type] = pos;
    }
    r.be32();  // toc crc
    StatisticsFile sf;
    if (toc.count(0)) {  // VALIDATION: partitioner class name (writeUTF) + fp chance
        Reader v(b.data() + toc[0], b.size() - toc[0]);
        uint16_t n = v.be16();
        bytes s = v.take(n);
        std::string pn((char*)s.data(), s.size());
        if (pn.find("ByteOrdered") != std::string::npos) sf.partitioner = Partitioner::BYTE_ORDERED;
    }
    if (!toc.count(3)) throw std::runtime_error("no HEADER component");
    {
        Reader h(b.data() + toc[3], b.size() - toc[3]);
        sf.header.stats.min_ts = (int64_t)read_unsigned_vint(h) + TIMESTAMP_EPOCH;
        sf.header.stats.min_ldt = (int64_t)(int32_t)(uint32_t)read_unsigned_vint(h) + DELETION_TIME_EPOCH;
        sf.header.stats.min_ttl = (int32_t)(uint32_t)read_unsigned_vint(h) + 0;
        auto read_type = [&]() {
            size_t n = (size_t)read_unsigned_vint(h);
            bytes s = h.take(n);
            return cql_type_from_name(std::string((char*)s.data(), s.size()));
        };
        sf.header.key_type = read_type();
        size_t nct = (size_t)read_unsigned_vint(h);
        for (size_t i = 0; i < nct; i++) sf.header.clustering_types.push_back(read_type());
        for (auto* cols : {&sf.header.static_cols, &sf.header.regular_cols}) {
            size_t nc = (size_t)read_unsigned_vint(h);
            for (size_t i = 0; i < nc; i++) {
                size_t nn = (size_t)read_unsigned_vint(h);
                bytes name = h.take(nn);
                cols->push_back({name, read_type()});
            }
        }
    }
    if (toc.count(2)) {
        Reader s(b.data() + toc[2], b.size() - toc[2]);
        for (int hh = 0; hh < 2; hh++) {  // two EstimatedHistograms
            uint32_t n = s.be32();
            s.skip((size_t)n * 16);
        }
        s.skip(12);  // CommitLogPosition
        sf.mins.min_timestamp = (int64_t)s.be64();
        sf.mins.max_timestamp = (int64_t)s.be64();
        sf.mins.min_ldt = ldt_to_long(s.be32());
        sf.mins.max_ldt = ldt_to_long(s.be32());
        sf.mins.min_ttl = (int32_t)s.be32();
        sf.mins.max_ttl = (int32_t)s.be32();
        // (rest of STATS not needed by the reader)
    }
    return sf;
}

static Partition read_partition(Reader& r, const Header& h,
                                std::vector<uint64_t>* item_offsets = nullptr) {
    Partition p;
    uint16_t klen = r.be16();
    p.key = r.take(klen);
    // token set by caller per partitioner
    p.del = read_deletion_time(r);
    if (h.has_static()) {
        uint8_t flags = r.u8();
        if (!(flags & F_EXTENSION)) throw std::runtime_error("expected static row extension flag");
        uint8_t xflags = r.u8();
        if (!(xflags & XF_IS_STATIC)) throw std::runtime_error("expected static row");
        p.static_row = read_row(r, h, flags, true);
    }
    while (true) {
        uint64_t item_pos = r.pos;
        uint8_t flags = r.u8();
        if (flags & F_END_OF_PARTITION) break;
        if (item_offsets) item_offsets->push_back(item_pos);
        Unfiltered u;
        if (flags & F_IS_MARKER) {
            u.kind = Unfiltered::MARKER;
            Marker& m = u.marker;
            m.kind = (BoundKind)r.u8();
            uint16_t nvals = r.be16();
            m.values = read_clustering_values(r, nvals, h.clustering_types);
            read_unsigned_vint(r);  // marker body size
            read_unsigned_vint(r);  // prev size
            if (m.boundary()) { m.end_dt = h.r_dt(r); m.start_dt = h.r_dt(r); }
            else m.end_dt = h.r_dt(r);
        } else {
            u.kind = Unfiltered::ROW;
            if (flags & F_EXTENSION) throw std::runtime_error("extended row flags unsupported");
            u.row = read_row(r, h, flags, false);
        }
        p.items.push_back(std::move(u));
    }
    return p;
}

ScrubResult scrub_sstable(const std::string& inbase, const std::string& outbase) {
    bytes ci_b = read_file(inbase + "-CompressionInfo.db");
    bytes data_b = read_file(inbase + "-Data.db");
    bytes stats_b = read_file(inbase + "-Statistics.db");
    bytes index_b = read_file(inbase + "-Index.db");
    CompressionInfo ci = read_compression_info(ci_b);
    StatisticsFile sf = read_statistics(stats_b);
    // Index.db -> partition positions
    std::vector<uint64_t> positions;
    {
        Reader r(index_b);
        while (!r.eof()) {
            uint16_t klen = r.be16();
            r.skip(klen);
            positions.push_back(read_unsigned_vint(r));
            uint64_t promoted = read_unsigned_vint(r);
            r.skip(promoted);
        }
        positions.push_back(ci.data_len);
    }
    // tolerant decompress with per-chunk bad flags
    size_t n_chunks = ci.offsets.size();
    std::vector<uint8_t> bad(n_chunks, 0);
    bytes raw(ci.data_len, 0);
    for (size_t i = 0; i < n_chunks; i++) {
        uint64_t off = ci.offsets[i];
        uint64_t end = (i + 1 < n_chunks) ? ci.offsets[i + 1] : data_b.size();
        uint64_t want = std::min<uint64_t>(ci.params.chunk_len, ci.data_len - i * (uint64_t)ci.params.chunk_len);
        if (end < off + 8) { bad[i] = 1; continue; }
        uint64_t comp_len = end - off - 4;
        uint32_t crc_stored = 0;
        for (int k = 0; k < 4; k++) crc_stored = (crc_stored << 8) | data_b[off + comp_len + k];
        if (crc_stored != crc32(data_b.data() + off, comp_len)) { bad[i] = 1; continue; }
        if (ci.params.algo == Compressor::SNAPPY) {
            size_t ulen = 0;
            if (!snappy_ref_uncompressed_length((const char*)data_b.data() + off, comp_len, &ulen) ||
                ulen != want ||
                !snappy_ref_uncompress((const char*)data_b.data() + off, comp_len,
                                       (char*)raw.data() + i * (uint64_t)ci.params.chunk_len))
                bad[i] = 1;
            continue;
        }
        uint32_t ulen = data_b[off] | (data_b[off + 1] << 8) | (data_b[off + 2] << 16) |
                        ((uint32_t)data_b[off + 3] << 24);
        if (ulen != want) { bad[i] = 1; continue; }
        int got = LZ4_decompress_safe((const char*)data_b.data() + off + 4,
                                      (char*)raw.data() + i * (uint64_t)ci.params.chunk_len,
                                      (int)(comp_len - 4), (int)ulen);
        if (got != (int)ulen) bad[i] = 1;
    }
    SSTable t;
    t.generation = 1;
    t.comp = ci.params;
    t.header = sf.header;  // rows DECODE with the input's HEADER mins
    uint64_t dropped = 0;
    for (size_t i = 0; i + 1 < positions.size(); i++) {
        uint64_t c0 = positions[i] / ci.params.chunk_len;
        uint64_t c1 = (positions[i + 1] + ci.params.chunk_len - 1) / ci.params.chunk_len;
        bool ok = true;
        for (uint64_t c = c0; c < c1 && c < n_chunks; c++)
            if (bad[c]) ok = false;
        if (!ok) { dropped++; continue; }
        bytes slice(raw.begin() + positions[i], raw.begin() + positions[i + 1]);
        Reader r(slice);
        Partition p = read_partition(r, t.header);
        p.token = sf.partitioner == Partitioner::MURMUR3
                      ? murmur3_token(p.key.data(), p.key.size()) : 0;
        t.parts.push_back(std::move(p));
    }
    // the REWRITE declares StatsMetadata mins as its header mins (what a
    // 1-input compaction would take) — set them only AFTER parsing: on a
    // compaction OUTPUT they can differ from the input's header mins, and
    // decoding with the writer's mins shifted every delta by a constant
    // (fuzz-caught: scrub of a TTL compaction output)
    t.header.stats.min_ts = sf.mins.min_timestamp == INT64_MIN ? TIMESTAMP_EPOCH : sf.mins.min_timestamp;
    t.header.stats.min_ldt = sf.mins.min_ldt == INT64_MAX ? DELETION_TIME_EPOCH : sf.mins.min_ldt;
    t.header.stats.min_ttl = sf.mins.min_ttl == INT32_MAX ? 0 : sf.mins.min_ttl;
    WriterOut w = write_sstable(t);
    write_components(w, outbase);
    ScrubResult sr;
    sr.kept = t.parts.size();
    sr.dropped = dropped;
    return sr;
}


Partition read_one_partition(const bytes& raw, uint64_t offset, const Header& h,
                             std::vector<uint64_t>* item_offsets) {
    Reader r(raw);
    r.skip(offset);
    return read_partition(r, h, item_offsets);
}

static bool file_exists_(const std::string& p2) {
    FILE* f = fopen(p2.c_str(), "rb");
    if (f) fclose(f);
    return f != nullptr;
}

SSTable read_sstable(const std::string& base, bool keep_raw) {
    SSTable t;
    t.bti = file_exists_(base + "-Partitions.db") && !file_exists_(base + "-Index.db");
    bytes ci_b = read_file(base + "-CompressionInfo.db");
    bytes data_b = read_file(base + "-Data.db");
    bytes stats_b = read_file(base + "-Statistics.db");
    CompressionInfo ci = read_compression_info(ci_b);
    t.comp = ci.params;
    StatisticsFile sf = read_statistics(stats_b);
    t.header = sf.header;
    t.stats = sf.mins;
    t.partitioner = sf.partitioner;
    bytes raw = decompress_data(data_b, ci);
    Reader r(raw);
    while (!r.eof()) {
        Partition p = read_partition(r, t.header);
        // token: Murmur3 normalize(hash[0]); BYTE_ORDERED orders by raw key
        // bytes — token 0 makes compare_decorated_key fall through to bytes.
        p.token = t.partitioner == Partitioner::MURMUR3 ? murmur3_token(p.key.data(), p.key.size()) : 0;
        t.parts.push_back(std::move(p));
    }
    // generation from ".../<ver>-<id>-big": parse numeric id if present
    {
        size_t p2 = base.rfind('/');
        std::string name = p2 == std::string::npos ? base : base.substr(p2 + 1);
        size_t a = name.find('-'), bpos = name.find('-', a + 1);
        if (a != std::string::npos && bpos != std::string::npos) {
            try { t.generation = std::stoull(name.substr(a + 1, bpos - a - 1)); } catch (...) { t.generation = 0; }
        }
    }
    if (keep_raw) { t.raw_data_uncompressed = std::move(raw); t.raw_statistics = std::move(stats_b); }
    return t;
}


// ---- memtable dump (flush-parity interchange) ----------------------------
// Serializes the in-memory SSTable (the logical memtable content) into a
// flat little-endian file that the GPU engine's gpuc_flush_table parity
// tests re-marshal through the C ABI. Test infrastructure: the format is
// ours (not a Cassandra component); the parity anchor is that
// gpuc_flush_table(memdump of t) must reproduce write_sstable(t) byte-
// for-byte.
namespace {
void md_u8(bytes& o, uint8_t v) { o.push_back(v); }
void md_u32(bytes& o, uint32_t v) { for (int i = 0; i < 4; i++) o.push_back((uint8_t)(v >> (8 * i))); }
void md_i32(bytes& o, int32_t v) { md_u32(o, (uint32_t)v); }
void md_u64(bytes& o, uint64_t v) { for (int i = 0; i < 8; i++) o.push_back((uint8_t)(v >> (8 * i))); }
void md_i64(bytes& o, int64_t v) { md_u64(o, (uint64_t)v); }
void md_blob(bytes& o, const bytes& b) { md_u32(o, (uint32_t)b.size()); o.insert(o.end(), b.begin(), b.end()); }
void md_cell(bytes& o, const Cell& c) {
    uint8_t cf = 1;                              // CELLF_PRESENT
    if (!c.tombstone()) cf |= 2;                 // CELLF_HAS_VALUE
    if (c.expiring()) cf |= 4;                   // CELLF_EXPIRING
    md_u8(o, cf);
    md_i64(o, c.ts);
    md_u32(o, c.ldt);
    md_i32(o, c.ttl);
    md_blob(o, c.value);
}
void md_clustering(bytes& o, const Clustering& c) {
    md_u8(o, (uint8_t)c.size());
    for (const auto& v : c) {
        md_u8(o, (uint8_t)v.state);
        md_blob(o, v.v);
    }
}
void md_row_head(bytes& o, const Row& r) {
    uint8_t f = 1;                               // PF_HAS_ROW analog
    if (!r.live.empty()) f |= 2;                 // PF_LIVE_TS
    if (!r.del.live()) f |= 4;                   // PF_ROW_DEL
    md_u8(o, f);
    md_i64(o, r.live.ts);
    md_i32(o, r.live.ttl);
    md_i64(o, r.live.let);
    md_i64(o, r.del.mfda);
    md_u32(o, r.del.ldt);
}
}  // namespace

void write_memdump(const SSTable& t, const std::string& path) {
    bytes o;
    o.insert(o.end(), {'G', 'M', 'D', '2'});
    md_u8(o, (uint8_t)t.header.key_type);
    md_u32(o, (uint32_t)t.header.clustering_types.size());
    for (CqlType ct : t.header.clustering_types) md_u8(o, (uint8_t)ct);
    md_u32(o, (uint32_t)t.header.static_cols.size());
    for (auto& [nm, ct] : t.header.static_cols) { md_blob(o, nm); md_u8(o, (uint8_t)ct); }
    md_u32(o, (uint32_t)t.header.regular_cols.size());
    for (auto& [nm, ct] : t.header.regular_cols) { md_blob(o, nm); md_u8(o, (uint8_t)ct); }
    md_u8(o, t.comp.algo == Compressor::SNAPPY ? 1 : 0);
    md_u8(o, t.bti ? 1 : 0);
    md_u32(o, t.column_index_size);
    // header EncodingStats: a flush CALLER provides these (the reference's
    // memtable passes its collected stats to SerializationHeader.make); a
    // compaction output's header mins come from its INPUT headers and are
    // not derivable from the data
    md_i64(o, t.header.stats.min_ts);
    md_i64(o, t.header.stats.min_ldt);
    md_i32(o, t.header.stats.min_ttl);
    md_u64(o, t.parts.size());
    const auto& regs = t.header.regular_cols;
    for (const Partition& p : t.parts) {
        md_u32(o, (uint32_t)p.key.size());
        o.insert(o.end(), p.key.begin(), p.key.end());
        md_i64(o, p.del.mfda);
        md_u32(o, p.del.ldt);
        if (!t.header.static_cols.empty()) {
            const Row& sr = p.static_row;
            if (row_is_empty(sr)) {
                md_u8(o, 0);
            } else {
                md_row_head(o, sr);
                for (size_t i = 0; i < t.header.static_cols.size(); i++) {
                    if (i < sr.cells.size() && sr.cells[i]) md_cell(o, *sr.cells[i]);
                    else md_u8(o, 0);
                }
            }
        }
        md_u32(o, (uint32_t)p.items.size());
        for (const Unfiltered& u : p.items) {
            if (u.kind == Unfiltered::ROW) {
                md_u8(o, 0);
                md_clustering(o, u.row.clustering);
                md_row_head(o, u.row);
                for (size_t i = 0; i < regs.size(); i++) {
                    if (regs[i].second == CqlType::MAP_BB) {
                        const auto& cd = i < u.row.complex.size() ? u.row.complex[i]
                                                                  : std::optional<ComplexData>{};
                        if (!cd) { md_u8(o, 0); continue; }
                        md_u8(o, 1);
                        md_i64(o, cd->del.mfda);
                        md_u32(o, cd->del.ldt);
                        md_u32(o, (uint32_t)cd->cells.size());
                        for (const Cell& c : cd->cells) {
                            md_cell(o, c);
                            md_blob(o, c.path);
                        }
                    } else if (i < u.row.cells.size() && u.row.cells[i]) {
                        md_cell(o, *u.row.cells[i]);
                    } else {
                        md_u8(o, 0);
                    }
                }
            } else {
                md_u8(o, 1);
                md_u8(o, (uint8_t)u.marker.kind);
                md_clustering(o, u.marker.values);
                md_i64(o, u.marker.end_dt.mfda);
                md_u32(o, u.marker.end_dt.ldt);
                md_i64(o, u.marker.start_dt.mfda);
                md_u32(o, u.marker.start_dt.ldt);
            }
        }
    }
    write_file(path, o);
}

}  // namespace oracle
