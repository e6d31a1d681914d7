// PRODUCT — host side of libcassandra_gpucompact: file I/O, sstable metadata
// (CompressionInfo/Statistics/Summary/TOC/Digest) assembly, and the HIP
// pipeline orchestration. Compiled as one TU with kernels.hip by hipcc.
//
// The host does NO data-path compute: decompress/decode/merge/reconcile/
// purge/serialize/compress/CRC/bloom all run in the kernels. There is no CPU
// fallback — gpuc_compact fails with GPUC_ERR_NO_GPU when no device exists.
#include "kernels.hip"
#include "kernels_rows.hip"
#include "bti_core.h"
#include "../../include/gpucompact.h"

#include <algorithm>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <map>
#include <set>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <future>
#include <atomic>
#include <fcntl.h>
#include <unistd.h>
#include <vector>

namespace gpuc {

using bytes = std::vector<uint8_t>;

#define HIP_CHECK(x)                                                              \
    do {                                                                          \
        hipError_t _e = (x);                                                      \
        if (_e != hipSuccess)                                                     \
            throw std::runtime_error(std::string("HIP error: ") +                 \
                                     hipGetErrorString(_e) + " at " #x);          \
    } while (0)

// ---------------------------------------------------------------------------
// host byte helpers
// ---------------------------------------------------------------------------
static void put_be16(bytes& o, uint16_t v) { o.push_back(v >> 8); o.push_back((uint8_t)v); }
static void put_be32(bytes& o, uint32_t v) { for (int i = 3; i >= 0; i--) o.push_back((uint8_t)(v >> (8 * i))); }
static void put_be64(bytes& o, uint64_t v) { for (int i = 7; i >= 0; i--) o.push_back((uint8_t)(v >> (8 * i))); }
static void put_uvint(bytes& o, uint64_t v) {
    uint8_t tmp[9];
    int n = uvint_put(tmp, v);
    o.insert(o.end(), tmp, tmp + n);
}

struct HReader {
    const uint8_t* p;
    size_t len, pos = 0;
    explicit HReader(const bytes& b) : p(b.data()), len(b.size()) {}
    void need(size_t n) const { if (pos + n > len) throw std::runtime_error("short read in component"); }
    uint8_t u8() { need(1); return p[pos++]; }
    uint16_t be16() { need(2); uint16_t v = ((uint16_t)p[pos] << 8) | p[pos + 1]; pos += 2; return v; }
    uint32_t be32() { need(4); uint32_t v = 0; for (int i = 0; i < 4; i++) v = (v << 8) | p[pos + i]; pos += 4; return v; }
    uint64_t be64() { need(8); uint64_t v = 0; for (int i = 0; i < 8; i++) v = (v << 8) | p[pos + i]; pos += 8; return v; }
    bytes take(size_t n) { need(n); bytes b(p + pos, p + pos + n); pos += n; return b; }
    void skip(size_t n) { need(n); pos += n; }
    uint64_t uvint() {
        need(1);
        uint64_t q = pos;
        uint64_t v = uvint_get(p, &q);
        if (q > len) throw std::runtime_error("vint overrun");
        pos = q;
        return v;
    }
};

// grow-only pinned staging buffers, reused across calls (hipHostMalloc is
// expensive; a compaction service reuses its staging arenas the same way)
struct PinnedSlot {
    void* p = nullptr;
    size_t cap = 0;
    void* get(size_t n) {
        if (n > cap) {
            if (p) (void)hipHostFree(p);
            size_t want = n + n / 4;
            if (hipHostMalloc(&p, want) != hipSuccess) { p = nullptr; cap = 0; return nullptr; }
            cap = want;
        }
        return p;
    }
};
static PinnedSlot g_pin_in[2][64], g_pin_out[2][4];

// GPUC_TRACE=1: wall-clock checkpoints on stderr (host-gap hunting)
static bool g_trace = getenv("GPUC_TRACE") != nullptr;
// validation mode: set by gpuc_validate around its gpuc_compact-shaped run
static thread_local const char* g_validate_out = nullptr;
static thread_local uint64_t g_validate_count = 0;
static double g_trace_t0 = 0;
static double trace_wall() {
    struct timespec ts; clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec * 1e3 + ts.tv_nsec / 1e6;
}
static void TR(const char* tag) {
    if (!g_trace) return;
    double t = trace_wall();
    fprintf(stderr, "[gpuc %8.1f] %s\n", t - g_trace_t0, tag);
}


// threaded read of a whole file into (pinned) memory
static size_t file_size_of(const std::string& path) {
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) throw std::runtime_error("cannot open " + path);
    fseek(f, 0, SEEK_END);
    long n = ftell(f);
    fclose(f);
    return (size_t)n;
}
static void read_file_range(const std::string& path, uint8_t* dst, uint64_t off, size_t n,
                            int nthreads = 3) {
    if (n == 0) return;
    std::vector<std::thread> th;
    size_t per = (n + nthreads - 1) / nthreads;
    for (int t = 0; t < nthreads; t++) {
        size_t o = (size_t)t * per;
        if (o >= n) break;
        size_t len = std::min(per, n - o);
        th.emplace_back([&, o, len]() {
            int fd = open(path.c_str(), O_RDONLY);
            if (fd < 0) return;
            size_t done = 0;
            while (done < len) {
                ssize_t r = pread(fd, dst + o + done, len - done, (off_t)(off + o + done));
                if (r <= 0) break;
                done += (size_t)r;
            }
            close(fd);
        });
    }
    for (auto& x : th) x.join();
}

static void write_file_parallel(const std::string& path, const uint8_t* p, size_t n, int nthreads = 4) {
    // create + size, then threaded pwrite
    FILE* f = fopen(path.c_str(), "wb");
    if (!f) throw std::runtime_error("cannot create " + path);
    fclose(f);
    if (n == 0) return;
    if (truncate(path.c_str(), (off_t)n) != 0) throw std::runtime_error("truncate " + path);
    std::vector<std::thread> th;
    size_t per = (n + nthreads - 1) / nthreads;
    for (int t = 0; t < nthreads; t++) {
        size_t off = (size_t)t * per;
        if (off >= n) break;
        size_t len = std::min(per, n - off);
        th.emplace_back([&, off, len]() {
            int fd = open(path.c_str(), O_WRONLY);
            if (fd < 0) return;
            size_t done = 0;
            while (done < len) {
                ssize_t w = pwrite(fd, p + off + done, len - done, (off_t)(off + done));
                if (w <= 0) break;
                done += (size_t)w;
            }
            close(fd);
        });
    }
    for (auto& x : th) x.join();
}

static bytes read_file(const std::string& path) {
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) throw std::runtime_error("cannot open " + path);
    fseek(f, 0, SEEK_END);
    long n = ftell(f);
    fseek(f, 0, SEEK_SET);
    bytes b((size_t)n);
    if (n && fread(b.data(), 1, (size_t)n, f) != (size_t)n) { fclose(f); throw std::runtime_error("short read " + path); }
    fclose(f);
    return b;
}
static void write_file(const std::string& path, const uint8_t* p, size_t n) {
    FILE* f = fopen(path.c_str(), "wb");
    if (!f) throw std::runtime_error("cannot create " + path);
    if (n && fwrite(p, 1, n, f) != n) { fclose(f); throw std::runtime_error("short write " + path); }
    fclose(f);
}

// ---------------------------------------------------------------------------
// component parsing (input side)
// ---------------------------------------------------------------------------
struct HCompressionInfo {
    uint32_t chunk_len = 16384;
    uint32_t max_compressed = 0x7FFFFFFF;
    uint64_t data_len = 0;
    bool snappy = false;  // SnappyCompressor (else LZ4Compressor)
    std::vector<uint64_t> offsets;
};
static HCompressionInfo parse_compression_info(const bytes& b) {
    HReader r(b);
    uint16_t nlen = r.be16();
    bytes name = r.take(nlen);
    std::string algo((char*)name.data(), name.size());
    if (algo != "LZ4Compressor" && algo != "SnappyCompressor")
        throw std::runtime_error("unsupported compressor " + algo + " (GPU path: LZ4/Snappy)");
    uint32_t opts = r.be32();
    for (uint32_t i = 0; i < opts; i++) { r.take(r.be16()); r.take(r.be16()); }
    HCompressionInfo ci;
    ci.snappy = algo == "SnappyCompressor";
    ci.chunk_len = r.be32();
    ci.max_compressed = r.be32();
    ci.data_len = r.be64();
    uint32_t n = r.be32();
    ci.offsets.resize(n);
    for (uint32_t i = 0; i < n; i++) ci.offsets[i] = r.be64();
    if (ci.chunk_len != CHUNK_LEN) throw std::runtime_error("unsupported chunk_length (16 KiB only in round 1)");
    return ci;
}

struct HStatistics {
    // HEADER component
    int64_t hdr_min_ts = TIMESTAMP_EPOCH;
    int64_t hdr_min_ldt = DELETION_TIME_EPOCH;
    int32_t hdr_min_ttl = 0;
    std::string key_type;
    std::vector<std::string> clustering_types;
    std::vector<std::pair<bytes, std::string>> static_cols, regular_cols;
    // STATS mins (for SerializationHeader.make of the output)
    int64_t min_timestamp = 0, max_timestamp = 0;
    int64_t min_ldt = NO_DELETION_TIME, max_ldt = 0;
    int32_t min_ttl = 0, max_ttl = 0;
    std::string partitioner;
};
static HStatistics parse_statistics(const bytes& b) {
    HReader r(b);
    uint32_t count = r.be32();
    r.be32();
    std::map<uint32_t, uint32_t> toc;
    for (uint32_t i = 0; i < count; i++) {
        uint32_t t = r.be32();
        uint32_t pos = r.be32();
        toc[t] = pos;
    }
    r.be32();
    HStatistics st;
    if (toc.count(0)) {
        HReader v(b);
        v.pos = toc[0];
        uint16_t n = v.be16();
        bytes s = v.take(n);
        st.partitioner.assign((char*)s.data(), s.size());
    }
    if (!toc.count(3)) throw std::runtime_error("Statistics.db missing HEADER component");
    {
        HReader h(b);
        h.pos = toc[3];
        st.hdr_min_ts = (int64_t)h.uvint() + TIMESTAMP_EPOCH;
        st.hdr_min_ldt = (int64_t)(int32_t)(uint32_t)h.uvint() + DELETION_TIME_EPOCH;
        st.hdr_min_ttl = (int32_t)(uint32_t)h.uvint();
        auto rstr = [&]() { size_t n = (size_t)h.uvint(); bytes s = h.take(n); return std::string((char*)s.data(), s.size()); };
        st.key_type = rstr();
        size_t nct = (size_t)h.uvint();
        for (size_t i = 0; i < nct; i++) st.clustering_types.push_back(rstr());
        for (auto* cols : {&st.static_cols, &st.regular_cols}) {
            size_t nc = (size_t)h.uvint();
            for (size_t i = 0; i < nc; i++) {
                size_t nn = (size_t)h.uvint();
                bytes name = h.take(nn);
                cols->push_back({name, rstr()});
            }
        }
    }
    if (toc.count(2)) {
        HReader s(b);
        s.pos = toc[2];
        for (int hh = 0; hh < 2; hh++) { uint32_t n = s.be32(); s.skip((size_t)n * 16); }
        s.skip(12);
        st.min_timestamp = (int64_t)s.be64();
        st.max_timestamp = (int64_t)s.be64();
        st.min_ldt = ldt_long(s.be32());
        st.max_ldt = ldt_long(s.be32());
        st.min_ttl = (int32_t)s.be32();
        st.max_ttl = (int32_t)s.be32();
    }
    return st;
}

// Index.db -> partition positions (host thread; validates non-indexed entries)
static void parse_index_positions(const bytes& ib, uint64_t data_len,
                                  std::vector<uint64_t>& positions,
                                  std::vector<uint64_t>* entry_offs, std::string& err) {
    try {
        HReader r(ib);
        positions.reserve(ib.size() / 18 + 2);
        if (entry_offs) entry_offs->reserve(ib.size() / 18 + 2);
        while (r.pos < r.len) {
            if (entry_offs) entry_offs->push_back(r.pos);
            uint16_t klen = r.be16();
            r.skip(klen);
            uint64_t pos = r.uvint();
            uint64_t promoted = r.uvint();
            r.skip(promoted);
            if (pos >= data_len || (!positions.empty() && pos <= positions.back()))
                throw std::runtime_error("Index.db positions not increasing/in range");
            positions.push_back(pos);
        }
        positions.push_back(data_len);
    } catch (const std::exception& e) {
        err = e.what();
    }
}

// ---------------------------------------------------------------------------
// output component assembly (host; small metadata only — spec'd identically in
// oracle/src/sstable.cpp so GPU and oracle outputs are byte-identical)
// ---------------------------------------------------------------------------
static void put_type_str(bytes& o, const std::string& s) {
    put_uvint(o, s.size());
    o.insert(o.end(), s.begin(), s.end());
}
static std::vector<int64_t> est_hist_offsets(int size) {
    std::vector<int64_t> off;
    int64_t last = 1;
    off.push_back(1);
    for (int i = 1; i < size; i++) {
        int64_t next = (int64_t)llround((double)last * 1.2);
        if (next == last) next++;
        off.push_back(next);
        last = next;
    }
    return off;
}
static void put_est_hist(bytes& o, const std::vector<int64_t>& off, const uint64_t* buckets) {
    put_be32(o, (uint32_t)(off.size() + 1));
    for (size_t i = 0; i < off.size() + 1; i++) {
        put_be64(o, (uint64_t)off[i == 0 ? 0 : i - 1]);
        put_be64(o, buckets[i]);
    }
}
// HyperLogLogPlus(13, 25) of the partition-key hash2_64 values (clearspring
// stream-lib; MetadataCollector.java:180-183). SPARSE while under the
// 0.75*m threshold with no flagged (low-bits-zero) key: (h >>> 39) << 1,
// delta varints over the sorted set — this encoding byte-reproduces the
// reference's own oa fixtures (oracle test_compaction_hll_fixture_pin).
// NORMAL otherwise: 2^13 five-bit registers, six per big-endian 32-bit word.
static bytes hll_bytes(const std::vector<uint64_t>& key_hashes) {
    constexpr int P = 13, SP = 25;
    bytes h;
    put_be32(h, (uint32_t)-2);
    auto pv = [&](uint32_t v) { while (v >= 0x80) { h.push_back((uint8_t)(v | 0x80)); v >>= 7; } h.push_back((uint8_t)v); };
    pv(P); pv(SP);
    std::set<uint32_t> sparse;
    bool flagged = false;
    const uint32_t threshold = (uint32_t)((1u << P) * 3 / 4);
    for (uint64_t kh : key_hashes) {
        uint32_t sidx = (uint32_t)(kh >> (64 - SP));
        if ((sidx & ((1u << (SP - P)) - 1)) == 0) { flagged = true; break; }
        sparse.insert(sidx << 1);
        if (sparse.size() > threshold) break;
    }
    if (!flagged && sparse.size() <= threshold && key_hashes.size() <= threshold) {
        pv(1 /*SPARSE*/);
        pv((uint32_t)sparse.size());
        uint32_t prev = 0;
        for (uint32_t v : sparse) { pv(v - prev); prev = v; }
    } else {
        pv(0 /*NORMAL*/);
        std::vector<uint8_t> regs(1u << P, 0);
        for (uint64_t kh : key_hashes) {
            uint32_t idx = (uint32_t)(kh >> (64 - P));
            uint64_t w = (kh << P) | (1ull << (P - 1));
            uint8_t rho = (uint8_t)(__builtin_clzll(w) + 1);
            if (rho > regs[idx]) regs[idx] = rho;
        }
        uint32_t bits = (1u << P) / 6;
        uint32_t reg_ints = (bits % 32 == 0) ? bits : bits + 1;  // RegisterSet.getSizeForCount
        std::vector<uint32_t> M(reg_ints, 0);
        for (uint32_t i = 0; i < (1u << P); i++)
            M[i / 6] |= (uint32_t)regs[i] << (5 * (i % 6));
        pv(reg_ints * 4);
        for (uint32_t wv : M) put_be32(h, wv);
    }
    bytes o;
    put_be32(o, (uint32_t)h.size());
    o.insert(o.end(), h.begin(), h.end());
    return o;
}

struct OutMeta {
    // header (deltas) of the output sstable
    HeaderStats hs;
    std::string key_type;
    std::vector<std::string> ck_types;  // empty = no clustering columns
    std::vector<std::pair<bytes, std::string>> static_cols;
    std::vector<std::pair<bytes, std::string>> regular_cols;
    // collected stats
    int64_t min_timestamp, max_timestamp, min_ldt, max_ldt;
    int32_t min_ttl, max_ttl;
    uint64_t total_rows, total_cells;
    bool has_partition_deletions;
    bytes first_key, last_key;
    std::map<uint32_t, uint32_t> tomb_hist;
    double compression_ratio;
    uint64_t part_size_hist[156];
    uint64_t cells_hist[119];
    std::vector<uint64_t> key_hashes;  // hash2_64 per kept partition (HLL)
};

static bytes serialize_statistics_out(const OutMeta& m) {
    bytes validation;
    {
        std::string pn = "org.apache.cassandra.dht.Murmur3Partitioner";
        put_be16(validation, (uint16_t)pn.size());
        validation.insert(validation.end(), pn.begin(), pn.end());
        uint64_t fp;
        double fpv = 0.01;
        memcpy(&fp, &fpv, 8);
        put_be64(validation, fp);
    }
    bytes compaction = hll_bytes(m.key_hashes);
    bytes stats;
    {
        static const std::vector<int64_t> ps_off = est_hist_offsets(155);
        static const std::vector<int64_t> ch_off = est_hist_offsets(118);
        put_est_hist(stats, ps_off, m.part_size_hist);
        put_est_hist(stats, ch_off, m.cells_hist);
        put_be64(stats, (uint64_t)-1LL); put_be32(stats, 0);  // commitLogUpperBound NONE
        put_be64(stats, (uint64_t)m.min_timestamp);
        put_be64(stats, (uint64_t)m.max_timestamp);
        put_be32(stats, ldt_u32(m.min_ldt));
        put_be32(stats, ldt_u32(m.max_ldt));
        put_be32(stats, (uint32_t)m.min_ttl);
        put_be32(stats, (uint32_t)m.max_ttl);
        uint64_t cr; double crv = m.compression_ratio; memcpy(&cr, &crv, 8); put_be64(stats, cr);
        put_be32(stats, 100);  // TombstoneHistogram maxBinSize
        put_be32(stats, (uint32_t)m.tomb_hist.size());
        for (auto& [pt, cnt] : m.tomb_hist) { put_be64(stats, pt); put_be32(stats, cnt); }
        put_be32(stats, 0);               // sstableLevel
        put_be64(stats, 0);               // repairedAt
        // improvedMinMax: clustering type list + covered Slice (spec'd with the
        // oracle: BOTTOM..TOP bounds, no values)
        put_uvint(stats, m.ck_types.size());
        for (auto& t : m.ck_types) put_type_str(stats, t);
        stats.push_back(1); put_be16(stats, 0);  // Slice start: INCL_START, 0 values
        stats.push_back(6); put_be16(stats, 0);  // Slice end: INCL_END, 0 values
        stats.push_back(0);               // hasLegacyCounterShards
        put_be64(stats, m.total_cells);
        put_be64(stats, m.total_rows);
        put_be64(stats, (uint64_t)-1LL); put_be32(stats, 0);  // commitLogLowerBound NONE
        put_be32(stats, 0);               // commitLogIntervals empty
        stats.push_back(0);               // pendingRepair null
        stats.push_back(0);               // isTransient
        stats.push_back(0);               // originatingHostId null
        stats.push_back(m.has_partition_deletions ? 1 : 0);
        put_uvint(stats, m.first_key.size());
        stats.insert(stats.end(), m.first_key.begin(), m.first_key.end());
        put_uvint(stats, m.last_key.size());
        stats.insert(stats.end(), m.last_key.begin(), m.last_key.end());
        uint64_t ts; double tsc = 0.0; memcpy(&ts, &tsc, 8); put_be64(stats, ts);
    }
    bytes header;
    {
        put_uvint(header, (uint64_t)(m.hs.min_ts - TIMESTAMP_EPOCH));
        put_uvint(header, sext32(m.hs.min_ldt - DELETION_TIME_EPOCH));
        put_uvint(header, sext32(m.hs.min_ttl));
        put_type_str(header, m.key_type);
        put_uvint(header, m.ck_types.size());
        for (auto& t : m.ck_types) put_type_str(header, t);
        put_uvint(header, m.static_cols.size());
        for (auto& [name, t] : m.static_cols) {
            put_uvint(header, name.size());
            header.insert(header.end(), name.begin(), name.end());
            put_type_str(header, t);
        }
        put_uvint(header, m.regular_cols.size());
        for (auto& [name, t] : m.regular_cols) {
            put_uvint(header, name.size());
            header.insert(header.end(), name.begin(), name.end());
            put_type_str(header, t);
        }
    }
    const bytes* comps[4] = {&validation, &compaction, &stats, &header};
    bytes out;
    uint32_t crc = 0;
    auto crc_int = [](uint32_t c, uint32_t v) {
        uint8_t b[4] = {(uint8_t)(v >> 24), (uint8_t)(v >> 16), (uint8_t)(v >> 8), (uint8_t)v};
        return crc32_update_bitwise(c, b, 4);
    };
    put_be32(out, 4);
    crc = crc_int(0, 4);
    put_be32(out, crc);
    uint32_t pos = 4 + 8 * 4 + 2 * 4;
    for (int i = 0; i < 4; i++) {
        put_be32(out, i);
        crc = crc_int(crc, i);
        put_be32(out, pos);
        crc = crc_int(crc, pos);
        pos += comps[i]->size() + 4;
    }
    put_be32(out, crc);
    for (int i = 0; i < 4; i++) {
        out.insert(out.end(), comps[i]->begin(), comps[i]->end());
        put_be32(out, crc32_update_bitwise(0, comps[i]->data(), comps[i]->size()));
    }
    return out;
}

// bloom spec for fp=0.01 (BloomCalculations; identical table in the oracle)
struct HBloomSpec { int k; int buckets; };
static HBloomSpec bloom_spec_001() { return {5, 10}; }  // computeBloomSpec(20, 0.01)

// ---------------------------------------------------------------------------
// device helpers (scan, merge)
// ---------------------------------------------------------------------------
// caching device allocator: hipMalloc of multi-GB buffers costs 100+ ms, and
// every compaction call uses the same transient working set, so freed blocks
// are pooled by power-of-two size class and reused across calls.
class DevPool {
    std::mutex mu_;
    std::map<size_t, std::vector<void*>> free_;

public:
    static size_t cls(size_t n) {
        size_t c = 4096;
        while (c < n) c <<= 1;
        return c;
    }
    void* get(size_t n) {
        size_t c = cls(n);
        {
            std::lock_guard<std::mutex> g(mu_);
            auto it = free_.find(c);
            if (it != free_.end() && !it->second.empty()) {
                void* p = it->second.back();
                it->second.pop_back();
                return p;
            }
        }
        void* p = nullptr;
        if (hipMalloc(&p, c) != hipSuccess) {
            trim();
            HIP_CHECK(hipMalloc(&p, c));
        }
        return p;
    }
    void put(size_t n, void* p) {
        std::lock_guard<std::mutex> g(mu_);
        free_[cls(n)].push_back(p);
    }
    void trim() {
        std::lock_guard<std::mutex> g(mu_);
        for (auto& [c, v] : free_) {
            for (void* p : v) (void)hipFree(p);
            v.clear();
        }
    }
};
static DevPool g_dev_pool;

struct DevBuf {
    void* p = nullptr;
    size_t n = 0;
    void alloc(size_t bytes_) {
        free_();
        n = bytes_ ? bytes_ : 8;
        p = g_dev_pool.get(n);
    }
    void free_() {
        if (p) {
            g_dev_pool.put(n, p);
            p = nullptr;
        }
    }
    ~DevBuf() { free_(); }
    template <typename T> T* as() const { return (T*)p; }
};

// exclusive scan over u64; returns total
static uint64_t exscan_u64(uint64_t* d_data, uint64_t n, hipStream_t stream) {
    if (n == 0) return 0;
    uint64_t n_blocks = (n + SCAN_TILE - 1) / SCAN_TILE;
    DevBuf sums, out;
    sums.alloc(n_blocks * 8);
    out.alloc(n * 8);
    hipLaunchKernelGGL(k_scan_partial, dim3((uint32_t)n_blocks), dim3(SCAN_BLOCK), 0, stream,
                       d_data, out.as<uint64_t>(), sums.as<uint64_t>(), n);
    uint64_t total = 0;
    if (n_blocks > 1) {
        total = exscan_u64(sums.as<uint64_t>(), n_blocks, stream);
        hipLaunchKernelGGL(k_scan_add, dim3((uint32_t)n_blocks), dim3(SCAN_BLOCK), 0, stream,
                           out.as<uint64_t>(), sums.as<uint64_t>(), n);
    } else {
        uint64_t s;
        HIP_CHECK(hipMemcpyAsync(&s, sums.p, 8, hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        total = s;
    }
    if (n_blocks > 1) {
        // total = scan total of sums + last block handled inside recursion; read final element
        uint64_t last_off, last_val;
        HIP_CHECK(hipMemcpyAsync(&last_off, out.as<uint64_t>() + (n - 1), 8, hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipMemcpyAsync(&last_val, d_data + (n - 1), 8, hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        total = last_off + last_val;
    }
    HIP_CHECK(hipMemcpyAsync(d_data, out.p, n * 8, hipMemcpyDeviceToDevice, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    return total;
}

// sort MRec array (uniform-run merge rounds); returns pointer to sorted buffer
static MRec* merge_sort_recs(MRec* d_a, MRec* d_b, uint64_t n, hipStream_t stream,
                             const KeyLut& lut) {
    MRec* in = d_a;
    MRec* out = d_b;
    for (uint64_t run = 1; run < n; run *= 2) {
        if (run < 16) {
            uint64_t pairs = (n + run * 2 - 1) / (run * 2);
            hipLaunchKernelGGL(k_merge_small, dim3((uint32_t)((pairs + 255) / 256)), dim3(256), 0,
                               stream, in, out, n, run, lut);
        } else {
            uint64_t tiles = (n + MERGE_TILE - 1) / MERGE_TILE;
            hipLaunchKernelGGL(k_merge_uniform, dim3((uint32_t)((tiles + 255) / 256)), dim3(256), 0,
                               stream, in, out, n, run, lut);
        }
        std::swap(in, out);
    }
    return in;
}

// merge k pre-sorted runs (per-source) with pairwise rounds
static MRec* merge_sorted_runs(MRec* d_a, MRec* d_b, std::vector<uint64_t> runs /*boundaries, size k+1*/,
                               hipStream_t stream, const KeyLut& lut) {
    MRec* in = d_a;
    MRec* out = d_b;
    while (runs.size() > 2) {
        std::vector<MergePair> pairs;
        std::vector<uint64_t> next_runs;
        next_runs.push_back(0);
        uint64_t tile_beg = 0;
        for (size_t i = 0; i + 1 < runs.size(); i += 2) {
            MergePair p{};
            p.a_beg = runs[i];
            p.a_end = runs[i + 1];
            if (i + 2 < runs.size()) { p.b_beg = runs[i + 1]; p.b_end = runs[i + 2]; }
            else { p.b_beg = p.a_end; p.b_end = p.a_end; }
            p.out_beg = p.a_beg;
            p.tile_beg = tile_beg;
            uint64_t n_out = (p.a_end - p.a_beg) + (p.b_end - p.b_beg);
            tile_beg += (n_out + MERGE_TILE - 1) / MERGE_TILE;
            pairs.push_back(p);
            next_runs.push_back(p.b_end);
        }
        DevBuf d_pairs;
        d_pairs.alloc(pairs.size() * sizeof(MergePair));
        HIP_CHECK(hipMemcpyAsync(d_pairs.p, pairs.data(), pairs.size() * sizeof(MergePair),
                                 hipMemcpyHostToDevice, stream));
        uint64_t total_tiles = tile_beg;
        hipLaunchKernelGGL(k_merge_pairs, dim3((uint32_t)((total_tiles + 255) / 256)), dim3(256), 0,
                           stream, in, out, d_pairs.as<MergePair>(), (uint32_t)pairs.size(), total_tiles, lut);
        HIP_CHECK(hipStreamSynchronize(stream));
        std::swap(in, out);
        runs = next_runs;
    }
    return in;
}

}  // namespace gpuc

namespace gpuc {

static void init_outstats(DevBuf& d_stats, hipStream_t stream) {
    hipLaunchKernelGGL(k_init_outstats, dim3(1), dim3(64), 0, stream, d_stats.as<OutStats>());
}

static bool g_crc_tables_ready = false;
static void* g_crc256 = nullptr;  // 8x256 sliced CRC tables (first 256 = base)
static std::mutex g_crc_init_mu;
static uint32_t* g_snp_off = nullptr;   // snappy probe-offset table (device)
static uint32_t g_snp_off_n = 0;

static void ensure_crc_tables(hipStream_t stream) {
    std::lock_guard<std::mutex> crc_g(g_crc_init_mu);
    if (g_crc_tables_ready) return;
    hipLaunchKernelGGL(k_crc_init, dim3(1), dim3(256), 0, stream);
    // plain 256-entry table for the wave LZ4 kernels
    uint32_t tab[256];
    crc32_make_table(tab);
    static uint32_t tab8[8 * 256];
    crc32_make_table8(tab8);
    HIP_CHECK(hipMalloc(&g_crc256, sizeof(tab8)));
    HIP_CHECK(hipMemcpyAsync(g_crc256, tab8, sizeof(tab8), hipMemcpyHostToDevice, stream));
    {   // snappy probe offsets (snappy.cc skip heuristic: skip=32, inc=skip>>5)
        std::vector<uint32_t> off{0};
        uint32_t skip = 32, o = 0;
        while (o < (1u << 16) + 128) {
            uint32_t inc = skip >> 5;
            skip += inc;
            o += inc;
            off.push_back(o);
        }
        g_snp_off_n = (uint32_t)off.size();
        HIP_CHECK(hipMalloc(&g_snp_off, off.size() * 4));
        HIP_CHECK(hipMemcpy(g_snp_off, off.data(), off.size() * 4, hipMemcpyHostToDevice));
    }
    // verify the LDS same-address write-order rule the wave compressor relies on
    unsigned int* d_probe;
    HIP_CHECK(hipMalloc(&d_probe, 8));
    hipLaunchKernelGGL(k_probe_lds_order, dim3(1), dim3(WAVE), 0, stream, d_probe);
    unsigned int probe[2];
    HIP_CHECK(hipStreamSynchronize(stream));
    HIP_CHECK(hipMemcpy(probe, d_probe, 8, hipMemcpyDeviceToHost));
    (void)hipFree(d_probe);
    if (probe[0] != 63 || probe[1] != 0)
        throw std::runtime_error("LDS conflicting-write order is not highest-lane-wins on this device; "
                                 "wave LZ4 compressor would be incorrect (refusing to run)");
    g_crc_tables_ready = true;
}

// allocate the OutParts SoA for n entries
// allocate a StaticCols block for n partitions x n_static columns
struct StaticColsBuf {
    DevBuf flags, live_ts, live_ttl, live_let, rdel_mfda, rdel_ldt,
        cell_flags, cell_ts, cell_ldt, cell_ttl, val_addr, val_len;
    StaticCols st{};
    void alloc(uint64_t n, uint32_t n_static) {
        if (!n) n = 1;
        uint64_t ns = n * (n_static ? n_static : 1);
        flags.alloc(n); live_ts.alloc(n * 8); live_ttl.alloc(n * 4); live_let.alloc(n * 8);
        rdel_mfda.alloc(n * 8); rdel_ldt.alloc(n * 4);
        cell_flags.alloc(ns); cell_ts.alloc(ns * 8); cell_ldt.alloc(ns * 4);
        cell_ttl.alloc(ns * 4); val_addr.alloc(ns * 8); val_len.alloc(ns * 4);
        st = StaticCols{flags.as<uint8_t>(), live_ts.as<int64_t>(), live_ttl.as<int32_t>(),
                        live_let.as<int64_t>(), rdel_mfda.as<int64_t>(), rdel_ldt.as<uint32_t>(),
                        cell_flags.as<uint8_t>(), cell_ts.as<int64_t>(), cell_ldt.as<uint32_t>(),
                        cell_ttl.as<int32_t>(), val_addr.as<uint64_t>(), val_len.as<uint32_t>()};
    }
};

struct OutPartsBuf {
    DevBuf keypfx, key_addr, token, klen, pdel_mfda, pdel_ldt, row_base, row_count, keep;
    StaticColsBuf stb;
    OutParts op{};
    DevBuf merged_k;
    void alloc(uint64_t n, uint32_t n_static = 0) {
        keypfx.alloc(n * 8); key_addr.alloc(n * 8); token.alloc(n * 8); klen.alloc(n * 2);
        pdel_mfda.alloc(n * 8); pdel_ldt.alloc(n * 4);
        row_base.alloc(n * 8); row_count.alloc(n * 4); keep.alloc(n);
        merged_k.alloc(n);
        HIP_CHECK(hipMemset(merged_k.p, 0, n));
        stb.alloc(n_static ? n : 1, n_static);
        op = OutParts{keypfx.as<uint64_t>(), key_addr.as<uint64_t>(), token.as<int64_t>(),
                      stb.st, klen.as<uint16_t>(),
                      pdel_mfda.as<int64_t>(),
                      pdel_ldt.as<uint32_t>(), row_base.as<uint64_t>(), row_count.as<uint32_t>(),
                      keep.as<uint8_t>(), merged_k.as<uint8_t>()};
    }
};

// allocate an UnfCols arena for n unfiltereds
struct UnfColsBuf {
    DevBuf ck, rkind, flags, live_ts, live_ttl, live_let, rdel_mfda, rdel_ldt,
        start_mfda, start_ldt, cell_ts, cell_ldt, cell_ttl, val_addr, val_len, ck_addr, ck_len,
        cell_flags, ck_count;
    DevBuf cpx_del_mfda, cpx_del_ldt, cpx_start, cpx_count;          // per row
    DevBuf cx_ts, cx_ldt, cx_ttl, cx_flags, cx_pa, cx_pl, cx_va, cx_vl;  // cell arena
    UnfCols uc{};
    // cell arena sized AFTER the counting pass (arena capacity = total
    // complex cells); per-row arrays are allocated with the rows
    void alloc_cpx_arena(uint64_t cells) {
        if (!cells) cells = 1;
        cx_ts.alloc(cells * 8); cx_ldt.alloc(cells * 4); cx_ttl.alloc(cells * 4);
        cx_flags.alloc(cells); cx_pa.alloc(cells * 8); cx_pl.alloc(cells * 4);
        cx_va.alloc(cells * 8); cx_vl.alloc(cells * 4);
        uc.cpx = UnfCols::CpxCells{cx_ts.as<int64_t>(), cx_ldt.as<uint32_t>(), cx_ttl.as<int32_t>(),
                                   cx_flags.as<uint8_t>(), cx_pa.as<uint64_t>(), cx_pl.as<uint32_t>(),
                                   cx_va.as<uint64_t>(), cx_vl.as<uint32_t>()};
    }
    void alloc(uint64_t n, uint32_t n_cols, uint32_t n_ck, uint32_t n_cpx = 0) {
        if (!n) n = 1;
        uint64_t nc = n * n_cols;
        uint64_t nk = n * (n_ck ? n_ck : 1);
        ck.alloc(nk * 8); rkind.alloc(n); flags.alloc(n); live_ts.alloc(n * 8);
        live_ttl.alloc(n * 4); live_let.alloc(n * 8); rdel_mfda.alloc(n * 8);
        rdel_ldt.alloc(n * 4); start_mfda.alloc(n * 8); start_ldt.alloc(n * 4);
        cell_ts.alloc(nc * 8); cell_ldt.alloc(nc * 4); cell_ttl.alloc(nc * 4);
        val_addr.alloc(nc * 8); val_len.alloc(nc * 4); cell_flags.alloc(nc);
        ck_addr.alloc(nk * 8); ck_len.alloc(nk * 4); ck_count.alloc(n);
        if (n_cpx) {
            cpx_del_mfda.alloc(n * 8); cpx_del_ldt.alloc(n * 4);
            cpx_start.alloc(n * 8); cpx_count.alloc(n * 4);
        }
        uc = UnfCols{ck.as<uint64_t>(), rkind.as<uint8_t>(), flags.as<uint8_t>(),
                     live_ts.as<int64_t>(), live_ttl.as<int32_t>(), live_let.as<int64_t>(),
                     rdel_mfda.as<int64_t>(), rdel_ldt.as<uint32_t>(), start_mfda.as<int64_t>(),
                     start_ldt.as<uint32_t>(), cell_ts.as<int64_t>(), cell_ldt.as<uint32_t>(),
                     cell_ttl.as<int32_t>(), val_addr.as<uint64_t>(), val_len.as<uint32_t>(),
                     cell_flags.as<uint8_t>(),
                     ck_addr.as<uint64_t>(), ck_len.as<uint32_t>(), ck_count.as<uint8_t>()};
        if (n_cpx) {
            uc.cpx_del_mfda = cpx_del_mfda.as<int64_t>();
            uc.cpx_del_ldt = cpx_del_ldt.as<uint32_t>();
            uc.cpx_start = cpx_start.as<uint64_t>();
            uc.cpx_count = cpx_count.as<uint32_t>();
        }
    }
};

struct WriteDeviceOut {
    uint64_t uncompressed_len = 0;
    uint64_t compressed_len = 0;   // final Data.db size
    uint64_t partitions = 0, rows = 0, cells = 0;
    double ms_sizes = 0, ms_serialize = 0, ms_compress = 0, ms_d2h = 0, ms_io = 0;
};

// shared device->files writer: sizes/scan -> serialize -> compress -> gather ->
// D2H -> write all components. `st` is the device OutStats already filled by
// reconcile (or gen); meta_* give header/schema info.
// BTI (`da`) index post-pass: Partitions.db + Rows.db built on the host
// from the big-format Index.db IMAGE the writer kernels already produced
// (the image carries keys, data positions and the promoted-index blocks —
// firstName/lastName clusterings, offsets, end-open markers — and those
// blocks ARE the bti row-index blocks: rowIndexBlockSize ==
// column_index_size, BtiFormatPartitionWriter.java:52). Byte layout pinned
// against the reference's legacy_da fixtures through the oracle
// (write_sstable(bti)) and the GPU parity suite.
static void build_bti_from_index_image(const uint8_t* idx, uint64_t idx_len, uint64_t data_total,
                                       const std::vector<int32_t>& ck_w,
                                       bti::bytes* partitions_out, bti::bytes* rows_out) {
    using bti::bytes;
    uint64_t p = 0;
    bytes rows_file;
    std::vector<bti::BtiKeyEntry> pes;
    std::vector<bti::BtiRowIndexBlockSpec> pending_specs;
    std::vector<size_t> pending_idx;
    std::vector<uint64_t> part_positions;
    auto uvint = [&]() -> uint64_t {
        uint8_t first = idx[p++];
        int extra = 0;
        uint8_t x = first;
        while (x & 0x80) { extra++; x <<= 1; }
        uint64_t r = first & (uint8_t)(0xFFu >> extra);
        for (int i = 0; i < extra; i++) r = (r << 8) | idx[p++];
        return r;
    };
    auto svint = [&]() -> int64_t {
        uint64_t u = uvint();
        return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
    };
    struct Blk {
        bytes first_bc, last_bc;
        uint64_t offset = 0;
        bool has_open = false;
        int64_t open_m = 0;
        uint32_t open_l = 0;
    };
    auto parse_prefix = [&]() -> bytes {
        uint8_t kind = idx[p++];
        uint32_t nv;
        if (kind != 4) {
            nv = ((uint32_t)idx[p] << 8) | idx[p + 1];
            p += 2;
        } else {
            nv = (uint32_t)ck_w.size();
        }
        std::vector<std::pair<const uint8_t*, uint32_t>> comps;
        std::vector<int32_t> widths;
        if (nv) {
            (void)uvint();  // 32-batch null header (0: all present)
            for (uint32_t c = 0; c < nv; c++) {
                uint32_t len = ck_w[c] > 0 ? (uint32_t)ck_w[c] : (uint32_t)uvint();
                comps.push_back({idx + p, len});
                widths.push_back(ck_w[c]);
                p += len;
            }
        }
        return bti::byte_comparable_clustering(comps, widths, kind);
    };
    while (p < idx_len) {
        uint32_t klen = ((uint32_t)idx[p] << 8) | idx[p + 1];
        p += 2;
        const uint8_t* key = idx + p;
        p += klen;
        uint64_t pos = uvint();
        uint64_t promoted = uvint();
        int64_t idxpos = ~(int64_t)pos;
        if (promoted > 0) {
            uint64_t pend = p + promoted;
            (void)uvint();  // headerLength
            bool pdel_live = idx[p] == 0x80;
            int64_t pdm = 0;
            uint32_t pdl = 0;
            if (pdel_live) {
                p += 1;
            } else {
                uint64_t m2 = 0;
                for (int i = 0; i < 8; i++) m2 = (m2 << 8) | idx[p + i];
                pdm = (int64_t)m2;
                pdl = ((uint32_t)idx[p + 8] << 24) | ((uint32_t)idx[p + 9] << 16) |
                      ((uint32_t)idx[p + 10] << 8) | idx[p + 11];
                p += 12;
            }
            uint64_t nblocks = uvint();
            std::vector<Blk> blocks;
            for (uint64_t b = 0; b < nblocks; b++) {
                Blk bk;
                bk.first_bc = parse_prefix();
                bk.last_bc = parse_prefix();
                bk.offset = uvint();
                (void)svint();  // width - 64KiB (unused here)
                bk.has_open = idx[p++] != 0;
                if (bk.has_open) {
                    if (idx[p] == 0x80) {
                        bk.has_open = false;
                        p += 1;
                    } else {
                        uint64_t m3v = 0;
                        for (int i = 0; i < 8; i++) m3v = (m3v << 8) | idx[p + i];
                        bk.open_m = (int64_t)m3v;
                        bk.open_l = ((uint32_t)idx[p + 8] << 24) | ((uint32_t)idx[p + 9] << 16) |
                                    ((uint32_t)idx[p + 10] << 8) | idx[p + 11];
                        p += 12;
                    }
                }
                blocks.push_back(std::move(bk));
            }
            p = pend;  // skip the block-offsets u32 table
            bti::BtiRowIndexBlockSpec spec;
            spec.partition_key.assign(key, key + klen);
            spec.data_pos = pos;
            spec.block_count = blocks.size();
            spec.pdel_live = pdel_live;
            spec.pdel_mfda = pdm;
            spec.pdel_ldt = pdl;
            bytes prev_sep, prev_max;
            for (size_t b = 0; b < blocks.size(); b++) {
                bytes bkey;
                if (b > 0) bkey = bti::bti_separator_gt(blocks[b - 1].last_bc, blocks[b].first_bc);
                bool has_od = b > 0 && blocks[b - 1].has_open;
                int ob = 1;
                while (blocks[b].offset >> (8 * ob - 1)) ob++;  // signed SizedInts
                bti::BtiRowIndexBlockSpec::Entry e;
                e.prefix = bkey;
                e.pb = ob | (has_od ? 8 : 0);
                for (int i = ob - 1; i >= 0; i--)
                    e.payload.push_back((uint8_t)(blocks[b].offset >> (8 * i)));
                if (has_od) {
                    for (int i = 7; i >= 0; i--)
                        e.payload.push_back((uint8_t)((uint64_t)blocks[b - 1].open_m >> (8 * i)));
                    for (int i = 3; i >= 0; i--)
                        e.payload.push_back((uint8_t)(blocks[b - 1].open_l >> (8 * i)));
                }
                spec.entries.push_back(std::move(e));
                prev_sep = bkey;
                prev_max = blocks[b].last_bc;
            }
            // final nudge entry; its payload (partition length - 1, the
            // END_OF_PARTITION byte's offset) is patched once the next
            // partition's position is known
            size_t cm = 0;
            while (cm < prev_max.size() && cm < prev_sep.size() && prev_max[cm] == prev_sep[cm]) cm++;
            bti::BtiRowIndexBlockSpec::Entry fin;
            fin.prefix = bti::bti_nudge(prev_max, cm);
            spec.entries.push_back(std::move(fin));
            pending_specs.push_back(std::move(spec));
            pending_idx.push_back(pes.size());
        }
        bti::BtiKeyEntry e;
        int64_t tok = murmur3_token(key, klen);
        e.byte_comparable = bti::bti_byte_comparable_m3(tok, bytes(key, key + klen));
        e.raw_key.assign(key, key + klen);
        uint64_t h2[2];
        murmur3_128(key, klen, 0, h2);
        e.hash_bits = (uint8_t)h2[1];  // DecoratedKey.filterHashLowerBits
        e.idxpos = idxpos;
        pes.push_back(std::move(e));
        part_positions.push_back(pos);
    }
    for (size_t i = 0; i < pending_specs.size(); i++) {
        auto& spec = pending_specs[i];
        size_t pi = pending_idx[i];
        uint64_t next_pos = pi + 1 < part_positions.size() ? part_positions[pi + 1] : data_total;
        uint64_t endoff = next_pos - spec.data_pos - 1;
        auto& fin = spec.entries.back();
        int ob = 1;
        while (endoff >> (8 * ob - 1)) ob++;
        fin.pb = ob;
        fin.payload.clear();
        for (int b = ob - 1; b >= 0; b--) fin.payload.push_back((uint8_t)(endoff >> (8 * b)));
        pes[pi].idxpos = (int64_t)bti::append_bti_row_index(rows_file, spec);
    }
    *partitions_out = bti::write_bti_partitions(pes);
    *rows_out = std::move(rows_file);
}

static WriteDeviceOut write_sstable_device(const OutPartsBuf& opb, const UnfColsBuf& rows,
                                           uint64_t n_groups, SerParams2 sp, DevBuf& d_stats,
                                           DevBuf& d_tomb, uint32_t tomb_cap,
                                           const std::string& out_base,
                                           const std::string& key_type,
                                           const std::vector<std::string>& ck_types,
                                           const std::vector<std::pair<bytes, std::string>>& regular_cols,
                                           const std::vector<std::pair<bytes, std::string>>& static_cols,
                                           hipStream_t stream, int wslot = 0,
                                           bool snappy_out = false, bool bti_out = false) {
    WriteDeviceOut w;
    const uint32_t SLOT_STRIDE = snappy_out ? SNP_SLOT : LZ4_SLOT;
    TR("wsd: enter");
    static const std::vector<int64_t> ps_off_h = est_hist_offsets(155);
    static const std::vector<int64_t> ch_off_h = est_hist_offsets(118);
    DevBuf d_ps_off, d_ch_off;
    d_ps_off.alloc(ps_off_h.size() * 8);
    d_ch_off.alloc(ch_off_h.size() * 8);
    HIP_CHECK(hipMemcpyAsync(d_ps_off.p, ps_off_h.data(), ps_off_h.size() * 8, hipMemcpyHostToDevice, stream));
    HIP_CHECK(hipMemcpyAsync(d_ch_off.p, ch_off_h.data(), ch_off_h.size() * 8, hipMemcpyHostToDevice, stream));

    hipEvent_t ev0, ev1, ev2, ev3, ev4;
    HIP_CHECK(hipEventCreate(&ev0)); HIP_CHECK(hipEventCreate(&ev1)); HIP_CHECK(hipEventCreate(&ev2));
    HIP_CHECK(hipEventCreate(&ev3)); HIP_CHECK(hipEventCreate(&ev4));

    // ---- collect stats (needed before bloom sizing) ----
    {
        uint64_t blocks = (n_groups + 255) / 256;
        hipLaunchKernelGGL(k_collect_rows, dim3((uint32_t)blocks), dim3(256), 0, stream, opb.op,
                           rows.uc, n_groups, sp.sch.n_cols, d_stats.as<OutStats>(),
                           d_tomb.as<uint32_t>(), tomb_cap);
    }

    // ---- sizes + scans ----
    HIP_CHECK(hipEventRecord(ev0, stream));
    DevBuf d_psize, d_isize, d_nblocks, d_infsz;
    d_psize.alloc(n_groups * 8);
    d_isize.alloc(n_groups * 8);
    d_nblocks.alloc(n_groups * 4);
    d_infsz.alloc(n_groups * 8);
    {
        uint64_t blocks = (n_groups + 255) / 256;
        hipLaunchKernelGGL(k_sizes_rows, dim3((uint32_t)blocks), dim3(256), 0, stream, opb.op,
                           rows.uc, n_groups, sp, d_psize.as<uint64_t>(), d_isize.as<uint64_t>(),
                           d_nblocks.as<uint32_t>(), d_infsz.as<uint64_t>(),
                           d_stats.as<OutStats>(), d_ps_off.as<int64_t>(), (int32_t)ps_off_h.size(),
                           d_ch_off.as<int64_t>(), (int32_t)ch_off_h.size());
    }
    uint64_t total_unc = exscan_u64(d_psize.as<uint64_t>(), n_groups, stream);
    {
        uint64_t blocks = (n_groups + 255) / 256;
        hipLaunchKernelGGL(k_index_sizes_rows, dim3((uint32_t)blocks), dim3(256), 0, stream,
                           opb.op, n_groups, sp, d_psize.as<uint64_t>(), d_nblocks.as<uint32_t>(),
                           d_infsz.as<uint64_t>(), d_isize.as<uint64_t>());
    }
    uint64_t total_idx = exscan_u64(d_isize.as<uint64_t>(), n_groups, stream);

    OutStats hst;
    HIP_CHECK(hipStreamSynchronize(stream));
    TR("wsd: sizes synced");
    HIP_CHECK(hipMemcpy(&hst, d_stats.p, sizeof(OutStats), hipMemcpyDeviceToHost));
    w.partitions = hst.partitions_out;
    w.rows = hst.rows_out;
    w.cells = hst.total_cells;

    // ---- serialize (+ index + bloom) ----
    HBloomSpec bs = bloom_spec_001();
    uint64_t num_bits = w.partitions * (uint64_t)bs.buckets + 20;
    uint64_t words = num_bits ? ((num_bits - 1) >> 6) + 1 : 1;
    DevBuf d_out_data, d_out_index, d_bloom;
    d_out_data.alloc(total_unc);
    d_out_index.alloc(total_idx);
    d_bloom.alloc(words * 8);
    HIP_CHECK(hipMemsetAsync(d_bloom.p, 0, words * 8, stream));
    // ev1 AFTER the bloom memset: segmented serialize (cstream) gates on it
    HIP_CHECK(hipEventRecord(ev1, stream));
    // serialize in SEGMENTS INTERLEAVED with the compress slabs on the MAIN
    // stream (in-order launches are the only synchronization): compress of a
    // byte range starts once the segment covering it has been issued ahead
    // of it, so the drain receives early slabs ~7/8 of a serialize earlier.
    // NEVER a third stream, and nothing ahead of the drain ops on cstream:
    // both alternatives measured -30% whole-step
    // (profiles/r02_drain_regression.md).
    hipStream_t cstream;
    HIP_CHECK(hipStreamCreate(&cstream));
    int NSEG = n_groups >= 4096 && total_unc > (256ull << 20) ? 8 : 1;
    if (const char* e = getenv("GPUC_NSEG")) NSEG = std::max(1, std::min(64, atoi(e)));
    if (n_groups < 4096) NSEG = 1;
    std::vector<uint64_t> seg_g(NSEG + 1), seg_end_byte(NSEG);
    if (NSEG > 1) {
        // equal-GROUP segments; only the NSEG-1 boundary byte offsets come
        // back from d_psize (not the whole offset array)
        seg_g[0] = 0;
        seg_g[NSEG] = n_groups;
        for (int j = 1; j < NSEG; j++) seg_g[j] = n_groups * (uint64_t)j / NSEG;
        for (int j = 0; j < NSEG - 1; j++)
            HIP_CHECK(hipMemcpy(&seg_end_byte[j], d_psize.as<uint64_t>() + seg_g[j + 1], 8,
                                hipMemcpyDeviceToHost));
        seg_end_byte[NSEG - 1] = total_unc;
    } else {
        seg_g[0] = 0;
        seg_g[NSEG] = n_groups;
        seg_end_byte[NSEG - 1] = total_unc;
    }
    auto launch_serialize_seg = [&](int j) {
        uint32_t waves_per_block = 4;
        uint64_t gs = seg_g[j], ge = seg_g[j + 1];
        uint64_t blocks = (std::max<uint64_t>(ge - gs, 1) + waves_per_block - 1) / waves_per_block;
        if (ge > gs)
            hipLaunchKernelGGL(k_serialize_rows, dim3((uint32_t)blocks),
                               dim3(WAVE * waves_per_block), 0, stream, opb.op, rows.uc, ge,
                               sp, d_psize.as<uint64_t>(), d_isize.as<uint64_t>(),
                               d_nblocks.as<uint32_t>(), d_infsz.as<uint64_t>(),
                               d_out_data.as<uint8_t>(), d_out_index.as<uint8_t>(),
                               d_bloom.as<uint32_t>(), words * 64, bs.k, gs);
    };
    int seg_issued = 0;
    if (NSEG == 1) {
        launch_serialize_seg(0);
        seg_issued = 1;
        HIP_CHECK(hipEventRecord(ev2, stream));
    }

    // ---- compress + gather + D2H + write: slab-pipelined ----
    // Compress launches for all slabs are enqueued on `stream` back to back;
    // a second stream drains finished slabs (sizes -> local offsets -> gather
    // -> D2H into rotating pinned buffers) while later slabs still compress,
    // and host writer threads pwrite each drained slab into Data.db at its
    // final offset. Output I/O therefore rides under the compress kernel
    // instead of following it.
    uint32_t n_chunks = (uint32_t)((total_unc + CHUNK_LEN - 1) / CHUNK_LEN);
    const uint32_t SLAB = 32768;  // 512 MiB uncompressed per slab
    uint32_t n_slabs = n_chunks ? (n_chunks + SLAB - 1) / SLAB : 0;
    DevBuf d_slots, d_csize, d_ccrc;
    d_slots.alloc((uint64_t)n_chunks * SLOT_STRIDE + 16);
    d_csize.alloc((uint64_t)n_chunks * 4 + 16);
    d_ccrc.alloc((uint64_t)n_chunks * 4 + 16);
    // Index.db image is final right after serialize: drain it early on
    // cstream (NSEG == 1; segmented mode enqueues it after ev2 exists, below)
    uint8_t* h_index = (uint8_t*)g_pin_out[wslot][1].get(total_idx ? total_idx : 1);
    if (!h_index) throw std::runtime_error("pinned out alloc failed");
    if (NSEG == 1) {
        HIP_CHECK(hipStreamWaitEvent(cstream, ev2, 0));
        HIP_CHECK(hipMemcpyAsync(h_index, d_out_index.p, total_idx, hipMemcpyDeviceToHost, cstream));
    }

    std::vector<hipEvent_t> ev_c(n_slabs);
    for (uint32_t i = 0; i < n_slabs; i++) {
        uint32_t cb = i * SLAB, m = std::min(SLAB, n_chunks - cb);
        if (NSEG > 1) {
            // issue the serialize segments covering this slab's bytes AHEAD
            // of it on the same in-order stream (ordering IS the dependency)
            uint64_t slab_end = std::min<uint64_t>((uint64_t)(cb + m) * CHUNK_LEN, total_unc);
            while (seg_issued < NSEG && (seg_issued ? seg_end_byte[seg_issued - 1] : 0) < slab_end)
                launch_serialize_seg(seg_issued++);
            if (seg_issued == NSEG) {
                HIP_CHECK(hipEventRecord(ev2, stream));
                seg_issued++;  // record ev2 once
            }
        }
        // all kernel arguments shift uniformly per chunk, so a slab launch is
        // just base-offset pointers with a local chunk count
        if (snappy_out)
            hipLaunchKernelGGL(k_snappy_compress_chunks, dim3(m), dim3(WAVE), 0, stream,
                               d_out_data.as<uint8_t>() + (uint64_t)cb * CHUNK_LEN,
                               total_unc - (uint64_t)cb * CHUNK_LEN,
                               d_slots.as<uint8_t>() + (uint64_t)cb * SNP_SLOT,
                               d_csize.as<uint32_t>() + cb, d_ccrc.as<uint32_t>() + cb, m,
                               (const uint32_t*)g_crc256, g_snp_off, g_snp_off_n);
        else
            hipLaunchKernelGGL(k_lz4_compress_wave_t<false>, dim3(m), dim3(WAVE), 0, stream,
                           d_out_data.as<uint8_t>() + (uint64_t)cb * CHUNK_LEN,
                           total_unc - (uint64_t)cb * CHUNK_LEN,
                           d_slots.as<uint8_t>() + (uint64_t)cb * LZ4_SLOT,
                           d_csize.as<uint32_t>() + cb, d_ccrc.as<uint32_t>() + cb, m,
                           (const uint32_t*)g_crc256);
        HIP_CHECK(hipEventCreate(&ev_c[i]));
        HIP_CHECK(hipEventRecord(ev_c[i], stream));
    }
    if (NSEG > 1 && seg_issued <= NSEG) {
        while (seg_issued < NSEG) launch_serialize_seg(seg_issued++);
        HIP_CHECK(hipEventRecord(ev2, stream));
    }
    HIP_CHECK(hipEventRecord(ev3, stream));  // ev2..ev3: compress (+interleaved serialize) GPU time
    TR("wsd: compress issued");

    std::string data_path = out_base + "-Data.db";
    {   // create/truncate so writers can pwrite into it
        FILE* f = fopen(data_path.c_str(), "wb");
        if (!f) throw std::runtime_error("cannot create " + data_path);
        fclose(f);
    }
    const uint64_t worst_slab = (uint64_t)SLAB * ((uint64_t)SLOT_STRIDE + 4);
    const int NSLOTS = 4;
    uint8_t* h_slab0 = (uint8_t*)g_pin_out[wslot][0].get(worst_slab * NSLOTS);
    if (!h_slab0) throw std::runtime_error("pinned out alloc failed");
    DevBuf d_gat[2], d_foff[2];
    if (n_slabs) {
        d_gat[0].alloc(worst_slab); d_gat[1].alloc(worst_slab);
        d_foff[0].alloc((uint64_t)SLAB * 8); d_foff[1].alloc((uint64_t)SLAB * 8);
    }
    std::vector<uint32_t> cs(n_chunks);
    std::vector<uint64_t> foff_h[2];
    foff_h[0].resize(SLAB); foff_h[1].resize(SLAB);
    std::future<void> wfut[NSLOTS];
    double t_drain0 = 0, t_drain1 = 0;
    {
        struct timespec tsd; clock_gettime(CLOCK_MONOTONIC, &tsd);
        t_drain0 = tsd.tv_sec * 1e3 + tsd.tv_nsec / 1e6;
    }
    uint64_t file_off = 0;
    double tw_comp = 0, tw_d2h = 0, tw_wfut = 0;  // drain wait breakdown (GPUC_TRACE)
    auto wallm = []() {
        struct timespec t; clock_gettime(CLOCK_MONOTONIC, &t);
        return t.tv_sec * 1e3 + t.tv_nsec / 1e6;
    };
    for (uint32_t i = 0; i < n_slabs; i++) {
        uint32_t cb = i * SLAB, m = std::min(SLAB, n_chunks - cb);
        HIP_CHECK(hipStreamWaitEvent(cstream, ev_c[i], 0));
        HIP_CHECK(hipMemcpyAsync(cs.data() + cb, d_csize.as<uint32_t>() + cb, (uint64_t)m * 4,
                                 hipMemcpyDeviceToHost, cstream));
        double tq0 = wallm();
        HIP_CHECK(hipStreamSynchronize(cstream));
        tw_comp += wallm() - tq0;
        auto& fo = foff_h[i & 1];
        uint64_t acc = 0;
        for (uint32_t j = 0; j < m; j++) { fo[j] = acc; acc += cs[cb + j]; }
        uint64_t slab_bytes = acc + (uint64_t)m * 4;
        HIP_CHECK(hipMemcpyAsync(d_foff[i & 1].p, fo.data(), (uint64_t)m * 8,
                                 hipMemcpyHostToDevice, cstream));
        hipLaunchKernelGGL(k_chunk_gather, dim3(m), dim3(WAVE), 0, cstream,
                           d_slots.as<uint8_t>() + (uint64_t)cb * SLOT_STRIDE,
                           d_csize.as<uint32_t>() + cb, d_ccrc.as<uint32_t>() + cb,
                           d_foff[i & 1].as<uint64_t>(), d_gat[i & 1].as<uint8_t>(), m,
                           SLOT_STRIDE);
        int slot = (int)(i % NSLOTS);
        double tq1 = wallm();
        if (wfut[slot].valid()) wfut[slot].get();  // pinned buffer free again
        tw_wfut += wallm() - tq1;
        uint8_t* hbuf = h_slab0 + (uint64_t)slot * worst_slab;
        HIP_CHECK(hipMemcpyAsync(hbuf, d_gat[i & 1].p, slab_bytes, hipMemcpyDeviceToHost, cstream));
        double tq2 = wallm();
        HIP_CHECK(hipStreamSynchronize(cstream));
        tw_d2h += wallm() - tq2;
        uint64_t off0 = file_off;
        wfut[slot] = std::async(std::launch::async, [=]() {
            int nth = 4;  // page-cache pwrites peak at ~4 threads (tools/wbench.c matrix)
            std::vector<std::thread> th;
            size_t per = (slab_bytes + nth - 1) / nth;
            for (int t = 0; t < nth; t++) {
                size_t o = (size_t)t * per;
                if (o >= slab_bytes) break;
                size_t len = std::min<size_t>(per, slab_bytes - o);
                th.emplace_back([=]() {
                    int fd = open(data_path.c_str(), O_WRONLY);
                    if (fd < 0) return;
                    size_t done = 0;
                    while (done < len) {
                        ssize_t ww = pwrite(fd, hbuf + o + done, len - done, (off_t)(off0 + o + done));
                        if (ww <= 0) break;
                        done += (size_t)ww;
                    }
                    close(fd);
                });
            }
            for (auto& x : th) x.join();
        });
        file_off += slab_bytes;
    }
    {
        double tq3 = wallm();
        for (int sfin = 0; sfin < NSLOTS; sfin++)
            if (wfut[sfin].valid()) wfut[sfin].get();
        tw_wfut += wallm() - tq3;
    }
    if (g_trace)
        fprintf(stderr, "[drain] slabs=%u wait_compress=%.0fms wait_d2h=%.0fms wait_write=%.0fms\n",
                n_slabs, tw_comp, tw_d2h, tw_wfut);
    {
        struct timespec tsd; clock_gettime(CLOCK_MONOTONIC, &tsd);
        t_drain1 = tsd.tv_sec * 1e3 + tsd.tv_nsec / 1e6;
    }
    w.compressed_len = file_off;
    TR("wsd: drain done");
    if (NSEG > 1) {
        // segmented mode: the index image drains AFTER the data drain — an
        // earlier enqueue would gate the whole in-order cstream on ev2 (the
        // LAST serialize segment) and collapse the drain to a tail
        HIP_CHECK(hipStreamWaitEvent(cstream, ev2, 0));
        HIP_CHECK(hipMemcpyAsync(h_index, d_out_index.p, total_idx, hipMemcpyDeviceToHost, cstream));
        HIP_CHECK(hipStreamSynchronize(cstream));
    }
    HIP_CHECK(hipEventRecord(ev4, stream));

    {
        std::vector<uint8_t> h_bloom(words * 8);
        std::vector<uint32_t> h_crc(n_chunks);
        TR("wsd: tail vectors");
        HIP_CHECK(hipMemcpy(h_bloom.data(), d_bloom.p, words * 8, hipMemcpyDeviceToHost));
        TR("wsd: bloom d2h");
        if (n_chunks)
            HIP_CHECK(hipMemcpy(h_crc.data(), d_ccrc.p, (uint64_t)n_chunks * 4, hipMemcpyDeviceToHost));
        HIP_CHECK(hipStreamSynchronize(stream));
        HIP_CHECK(hipStreamSynchronize(cstream));
        TR("wsd: tail synced");
        // digest fold and the Summary.db index walk are independent of the
        // meta build: run all three concurrently (tail was serial: ~145 ms)
        static Crc32Combiner comb;
        std::future<uint32_t> fut_digest = std::async(std::launch::async, [&]() {
            int nth = n_chunks > 4096 ? 16 : 1;
            std::vector<uint32_t> pcrc(nth, 0);
            std::vector<uint64_t> plen(nth, 0);
            uint32_t per = (n_chunks + nth - 1) / nth;
            std::vector<std::thread> th;
            for (int t = 0; t < nth; t++) {
                th.emplace_back([&, t]() {
                    uint32_t c0 = t * per, c1 = std::min(n_chunks, (t + 1) * per);
                    uint32_t d = 0;
                    uint64_t l = 0;
                    for (uint32_t c = c0; c < c1; c++) {
                        d = comb.combine(d, h_crc[c], cs[c]);
                        uint8_t cb2[4] = {(uint8_t)(h_crc[c] >> 24), (uint8_t)(h_crc[c] >> 16),
                                          (uint8_t)(h_crc[c] >> 8), (uint8_t)h_crc[c]};
                        d = comb.combine(d, crc32_update_bitwise(0, cb2, 4), 4);
                        l += cs[c] + 4;
                    }
                    pcrc[t] = d;
                    plen[t] = l;
                });
            }
            for (auto& x : th) x.join();
            uint32_t dg = 0;
            for (int t = 0; t < nth; t++) dg = comb.combine(dg, pcrc[t], plen[t]);
            return dg;
        });

        std::future<bytes> fut_summary;
        if (!bti_out)
            fut_summary = std::async(std::launch::async, [&]() -> bytes {
            // IndexSummary at BASE_SAMPLING_LEVEL (IndexSummaryBuilder
            // with empty Downsampling start points: entries for keys
            // 0, 128, 256, ...), serialized per
            // IndexSummary.IndexSummarySerializer.serialize — BE header,
            // then the off-heap image in NATIVE (LE) order: rebased
            // int offsets, then key bytes + LE u64 Index.db position;
            // trailing first/last key with BE lengths
            // (SSTableReader.saveSummary). Fixture-pinned by the
            // oracle roundtrip of all four legacy_oa tables.
            const uint32_t MIN_INTERVAL = 128;
            bytes entries, s;
            std::vector<uint32_t> offs;
            const uint8_t *first_k = nullptr, *last_k = nullptr;
            uint32_t first_kl = 0, last_kl = 0;
            uint64_t q = 0, part_i = 0;
            while (q < total_idx) {
                uint64_t entry_off = q;
                uint32_t klen = ((uint32_t)h_index[q] << 8) | h_index[q + 1];
                const uint8_t* kp = h_index + q + 2;
                q += 2 + klen;
                if (!first_k) { first_k = kp; first_kl = klen; }
                last_k = kp; last_kl = klen;
                if (part_i % MIN_INTERVAL == 0) {
                    offs.push_back((uint32_t)entries.size());
                    entries.insert(entries.end(), kp, kp + klen);
                    for (int b = 0; b < 8; b++) entries.push_back((uint8_t)(entry_off >> (8 * b)));
                }
                // skip position + promoted-index payload vints
                auto skip_uvint = [&]() -> uint64_t {
                    uint8_t f2 = h_index[q++];
                    int extra = 0;
                    uint8_t x = f2;
                    while (x & 0x80) { extra++; x <<= 1; }
                    uint64_t r2 = f2 & (uint8_t)(0xFFu >> extra);
                    for (int i2 = 0; i2 < extra; i2++) r2 = (r2 << 8) | h_index[q++];
                    return r2;
                };
                (void)skip_uvint();
                q += skip_uvint();
                part_i++;
            }
            uint32_t cnt = (uint32_t)offs.size();
            put_be32(s, MIN_INTERVAL);
            put_be32(s, cnt);
            put_be64(s, 4ull * cnt + entries.size());
            put_be32(s, 128);
            put_be32(s, cnt);
            for (uint32_t o : offs) {
                uint32_t v = o + 4 * cnt;
                for (int b = 0; b < 4; b++) s.push_back((uint8_t)(v >> (8 * b)));
            }
            s.insert(s.end(), entries.begin(), entries.end());
            if (first_k) {
                put_be32(s, first_kl);
                s.insert(s.end(), first_k, first_k + first_kl);
                put_be32(s, last_kl);
                s.insert(s.end(), last_k, last_k + last_kl);
            }
            return s;
            });

        std::vector<uint32_t> tombs;
        if (hst.tomb_count) {
            uint64_t nt = std::min<uint64_t>(hst.tomb_count, tomb_cap);
            tombs.resize(nt);
            HIP_CHECK(hipMemcpy(tombs.data(), d_tomb.p, nt * 4, hipMemcpyDeviceToHost));
            if (hst.tomb_count > tomb_cap)
                throw std::runtime_error("tombstone list overflow (internal cap)");
        }
        OutMeta m{};
        {
            // per-partition key hashes for the COMPACTION HLL (device-hashed,
            // filtered to kept partitions on host)
            DevBuf d_kh;
            d_kh.alloc(n_groups * 8 + 8);
            uint32_t blocks = (uint32_t)((n_groups + 255) / 256);
            if (n_groups)
                hipLaunchKernelGGL(k_key_hash2, dim3(blocks), dim3(256), 0, stream, opb.op,
                                   n_groups, d_kh.as<uint64_t>());
            std::vector<uint64_t> kh(n_groups);
            std::vector<uint8_t> keep(n_groups);
            if (n_groups) {
                HIP_CHECK(hipStreamSynchronize(stream));
                HIP_CHECK(hipMemcpy(kh.data(), d_kh.p, n_groups * 8, hipMemcpyDeviceToHost));
                HIP_CHECK(hipMemcpy(keep.data(), opb.op.keep, n_groups, hipMemcpyDeviceToHost));
            }
            m.key_hashes.reserve(n_groups);
            for (uint64_t g2 = 0; g2 < n_groups; g2++)
                if (keep[g2]) m.key_hashes.push_back(kh[g2]);
        }
        m.hs = sp.hs;
        m.key_type = key_type;
        m.ck_types = ck_types;
        m.regular_cols = regular_cols;
        m.static_cols = static_cols;
        bool no_ts = hst.min_ts_flip == 0xFFFFFFFFFFFFFFFFULL;
        m.min_timestamp = no_ts ? 0 : (int64_t)(hst.min_ts_flip ^ 0x8000000000000000ULL);
        m.max_timestamp = no_ts ? 0 : (int64_t)(hst.max_ts_flip ^ 0x8000000000000000ULL);
        m.min_ldt = hst.min_ldt_flip == 0xFFFFFFFFFFFFFFFFULL ? NO_DELETION_TIME
                                                              : (int64_t)(hst.min_ldt_flip ^ 0x8000000000000000ULL);
        m.max_ldt = hst.max_ldt_flip == 0 ? 0 : (int64_t)(hst.max_ldt_flip ^ 0x8000000000000000ULL);
        m.min_ttl = hst.min_ttl == 0xFFFFFFFFu ? 0 : (int32_t)hst.min_ttl;
        m.max_ttl = (int32_t)hst.max_ttl;
        m.total_rows = hst.rows_out;
        m.total_cells = hst.total_cells;
        m.has_partition_deletions = hst.has_partition_deletions != 0;
        for (uint32_t tv : tombs) m.tomb_hist[tv]++;
        memcpy(m.part_size_hist, hst.part_size_hist, sizeof(m.part_size_hist));
        memcpy(m.cells_hist, hst.cells_hist, sizeof(m.cells_hist));
        m.compression_ratio = total_unc ? (double)(w.compressed_len - (uint64_t)n_chunks * 4) / (double)total_unc : -1.0;
        if (w.partitions) {
            uint64_t fg = hst.first_group, lg = hst.last_group;
            for (int which = 0; which < 2; which++) {
                uint64_t g = which ? lg : fg;
                uint64_t ka;
                uint16_t kl;
                HIP_CHECK(hipMemcpy(&ka, opb.op.key_addr + g, 8, hipMemcpyDeviceToHost));
                HIP_CHECK(hipMemcpy(&kl, opb.op.klen + g, 2, hipMemcpyDeviceToHost));
                bytes kb(kl);
                HIP_CHECK(hipMemcpy(kb.data(), (const void*)ka, kl, hipMemcpyDeviceToHost));
                (which ? m.last_key : m.first_key) = kb;
            }
        }
        TR("wsd: meta built");
        // Digest = CRC of the whole Data.db, folded from per-chunk CRCs in
        // the async above (CRC concatenation is associative over (crc, len))
        uint32_t digest = fut_digest.get();

        TR("wsd: digest done");
        struct timespec ts0, ts1;
        clock_gettime(CLOCK_MONOTONIC, &ts0);
        if (bti_out) {
            bti::bytes parts_img, rows_img;
            std::vector<int32_t> ckw_h2;
            for (auto& t2 : ck_types)
                ckw_h2.push_back(t2 == "org.apache.cassandra.db.marshal.LongType" ? 8
                                 : t2 == "org.apache.cassandra.db.marshal.Int32Type" ? 4
                                                                                     : -1);
            build_bti_from_index_image(h_index, total_idx, total_unc, ckw_h2,
                                       &parts_img, &rows_img);
            write_file(out_base + "-Partitions.db", parts_img.data(), parts_img.size());
            write_file(out_base + "-Rows.db", rows_img.data(), rows_img.size());
        } else {
            write_file_parallel(out_base + "-Index.db", h_index, total_idx, 4);
        }
        {
            bytes f;
            put_be32(f, (uint32_t)bs.k);
            put_be32(f, (uint32_t)words);
            f.insert(f.end(), h_bloom.begin(), h_bloom.end());
            write_file(out_base + "-Filter.db", f.data(), f.size());
        }
        {
            bytes ci;
            std::string algo = snappy_out ? "SnappyCompressor" : "LZ4Compressor";
            put_be16(ci, (uint16_t)algo.size());
            ci.insert(ci.end(), algo.begin(), algo.end());
            put_be32(ci, 0);
            put_be32(ci, CHUNK_LEN);
            put_be32(ci, 0x7FFFFFFF);
            put_be64(ci, total_unc);
            put_be32(ci, n_chunks);
            uint64_t acc2 = 0;
            for (uint32_t i = 0; i < n_chunks; i++) { put_be64(ci, acc2 + (uint64_t)i * 4); acc2 += cs[i]; }
            write_file(out_base + "-CompressionInfo.db", ci.data(), ci.size());
        }
        {
            std::string d = std::to_string(digest);
            write_file(out_base + "-Digest.crc32", (const uint8_t*)d.data(), d.size());
        }
        {
            bytes st = serialize_statistics_out(m);
            write_file(out_base + "-Statistics.db", st.data(), st.size());
        }
        {
            if (!bti_out) {
                bytes s = fut_summary.get();
                write_file(out_base + "-Summary.db", s.data(), s.size());
            }
            std::string toc = bti_out
                ? "Data.db\nStatistics.db\nDigest.crc32\nTOC.txt\nCompressionInfo.db\nFilter.db\nPartitions.db\nRows.db\n"
                : "Data.db\nStatistics.db\nDigest.crc32\nTOC.txt\nCompressionInfo.db\nFilter.db\nIndex.db\nSummary.db\n";
            write_file(out_base + "-TOC.txt", (const uint8_t*)toc.data(), toc.size());
        }
        clock_gettime(CLOCK_MONOTONIC, &ts1);
        TR("wsd: components written");
        // residual (non-overlapped) component writes + the Data.db drain tail
        w.ms_io = (ts1.tv_sec - ts0.tv_sec) * 1e3 + (ts1.tv_nsec - ts0.tv_nsec) / 1e6;
        w.ms_d2h = t_drain1 - t_drain0;  // slab drain wall (overlaps compress)
    }
    for (uint32_t i = 0; i < n_slabs; i++) (void)hipEventDestroy(ev_c[i]);

    HIP_CHECK(hipStreamDestroy(cstream));
    w.uncompressed_len = total_unc;
    float t01, t12, t23;
    HIP_CHECK(hipEventElapsedTime(&t01, ev0, ev1));
    HIP_CHECK(hipEventElapsedTime(&t12, ev1, ev2));
    HIP_CHECK(hipEventElapsedTime(&t23, ev2, ev3));
    w.ms_sizes = t01;
    w.ms_serialize = t12;
    w.ms_compress = t23;
    for (auto e : {ev0, ev1, ev2, ev3, ev4}) (void)hipEventDestroy(e);
    return w;
}

}  // namespace gpuc

// ---------------------------------------------------------------------------
// C ABI
// ---------------------------------------------------------------------------
using namespace gpuc;

extern "C" const char* gpuc_version(void) { return "cassandra_gpucompact 0.1 (gfx950)"; }

extern "C" int gpuc_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

static void set_err(char* dst, size_t cap, const std::string& msg) {
    if (!dst || !cap) return;
    snprintf(dst, cap, "%s", msg.c_str());
}

// marshal type string -> clustering component width (-1 variable)
static int32_t ck_type_width(const std::string& t) {
    if (t == "org.apache.cassandra.db.marshal.LongType") return 8;
    if (t == "org.apache.cassandra.db.marshal.Int32Type") return 4;
    if (t == "org.apache.cassandra.db.marshal.UTF8Type" ||
        t == "org.apache.cassandra.db.marshal.AsciiType" ||
        t == "org.apache.cassandra.db.marshal.BytesType")
        return -1;
    throw std::runtime_error("unsupported clustering type " + t);
}

struct CompactSetup {
    int k = 0;        // total ingested sstables (data + tombstone sources)
    int k_data = 0;   // first k_data are the compacted data inputs
    std::vector<std::string> in_bases;
    std::vector<bytes> index_data;
    std::vector<HCompressionInfo> cinfos;
    std::vector<HStatistics> stats;
    std::vector<uint64_t> generations;
    std::vector<std::vector<uint64_t>> positions;   // n_parts+1 absolute offsets
    std::vector<std::vector<uint64_t>> entry_offs;  // Index.db entry byte offsets
    std::vector<std::vector<bti::bytes>> bti_prefixes;  // da: trie separator per partition
    std::vector<size_t> comp_file_sz;
    std::vector<int32_t> col_fixed_h;    // SIMPLE regular columns only
    uint32_t n_cpx = 0;                  // one complex (map<blob,blob>) column, last
    bool counters = false;               // counter table (every column CounterColumnType)
    bool bti = false;                    // inputs are `da` (trie-indexed)
    std::vector<int32_t> ck_widths;      // per clustering column
    std::vector<int32_t> static_fixed_h; // per static column
    // unsharded fast path: whole-file Data.db reads started during index
    // parse as ordered STRIPES per source (compact_one adopts them when its
    // window covers the full file and overlaps H2D with the remaining reads)
    struct StripeRead {
        mutable std::vector<std::future<void>> th;  // one per stripe, in file order
        std::vector<uint64_t> off, len;
    };
    mutable std::vector<StripeRead> full_stripes;
    // file-ordered reader pool: stripes are queued (file 0 first) so the
    // FIRST sstable's bytes land early and its H2D+decompress pipeline
    // starts ~1/k into the read instead of after all files finish together
    mutable std::vector<std::thread> read_pool;
    std::vector<uint8_t*> full_pin;
    ~CompactSetup() {
        for (auto& sr : full_stripes)
            for (auto& t : sr.th)
                if (t.valid()) t.wait();
        for (auto& t : read_pool)
            if (t.joinable()) t.join();
    }
};

// Everything from data read through component write for ONE output sstable,
// restricted to the partition window pr[s] = [blo, bhi) per input (the full
// range in the unsharded case). wslot selects the pinned-arena set; shards
// run two at a time so front-phase kernels of one shard overlap the other
// shard's LDS-bound compress (they use disjoint CU resources).
struct cancelled_error : std::runtime_error {
    cancelled_error() : std::runtime_error("compaction cancelled (cancel_flag set)") {}
};
// cooperative cancellation poll (CompactionIterator.isStopRequested):
// called between pipeline phases of a task
static inline void check_cancel(const gpuc_job* job) {
    if (job->cancel_flag && *job->cancel_flag) throw cancelled_error();
}

static void compact_one(const gpuc_job* job, const CompactSetup& su,
                        const std::vector<std::pair<uint32_t, uint32_t>>& pr,
                        const std::string& out_base_str, int wslot,
                        gpuc_result* res, std::mutex& res_mu,
                        bool shard_exact = false, int64_t sh_lo = 0, int64_t sh_hi = 0) {
    auto wall = []() {
        struct timespec ts;
        clock_gettime(CLOCK_MONOTONIC, &ts);
        return ts.tv_sec * 1e3 + ts.tv_nsec / 1e6;
    };
    HIP_CHECK(hipSetDevice(job->device));
    hipStream_t stream;
    HIP_CHECK(hipStreamCreate(&stream));
    {
        int k = su.k;
        const auto& in_bases = su.in_bases;
        const auto& cinfos = su.cinfos;
        const auto& stats = su.stats;
        const auto& generations = su.generations;
        const auto& positions = su.positions;
        const auto& col_fixed_h = su.col_fixed_h;

        double t0 = wall();
        // per-input window: partition range -> chunk range -> compressed range
        std::vector<uint32_t> win_blo(k), win_n(k);
        std::vector<uint64_t> win_chunk_lo(k), win_comp_lo(k);
        std::vector<size_t> comp_sz(k, 0);        // compressed bytes in window
        std::vector<uint8_t*> comp_pin(k, nullptr);
        std::vector<uint64_t> data_lo(k), data_hi(k);  // uncompressed byte window
        for (int s = 0; s < k; s++) {
            uint32_t blo = pr[s].first, bhi = pr[s].second;
            win_blo[s] = blo;
            win_n[s] = bhi - blo;
            data_lo[s] = positions[s][blo];
            data_hi[s] = positions[s][bhi];
            uint64_t c_lo = data_lo[s] / CHUNK_LEN;
            uint64_t c_hi = (data_hi[s] + CHUNK_LEN - 1) / CHUNK_LEN;
            uint64_t n_chunks_file = su.cinfos[s].offsets.size();
            if (c_hi > n_chunks_file) c_hi = n_chunks_file;
            if (win_n[s] == 0) { c_lo = c_hi = 0; }
            win_chunk_lo[s] = c_lo;
            uint64_t comp_lo = c_lo < n_chunks_file ? cinfos[s].offsets[c_lo] : su.comp_file_sz[s];
            uint64_t comp_hi = c_hi < n_chunks_file ? cinfos[s].offsets[c_hi] : su.comp_file_sz[s];
            if (win_n[s] == 0) comp_lo = comp_hi = 0;
            win_comp_lo[s] = comp_lo;
            comp_sz[s] = comp_hi - comp_lo;
        }
        bool adopt = !su.full_pin.empty();
        std::vector<std::thread> data_readers(adopt ? 0 : k);
        for (int s = 0; s < k; s++) {
            if (adopt) {
                // setup already reads the whole file into the slot-0 arenas
                comp_pin[s] = su.full_pin[s] + win_comp_lo[s];
            } else {
                comp_pin[s] = (uint8_t*)g_pin_in[wslot][s].get(comp_sz[s] ? comp_sz[s] : 1);
                if (!comp_pin[s]) throw std::runtime_error("pinned alloc failed");
                data_readers[s] = std::thread([&, s] {
                    if (comp_sz[s])
                        read_file_range(in_bases[s] + "-Data.db", comp_pin[s], win_comp_lo[s],
                                        comp_sz[s], 3);
                });
            }
        }
        {
            std::lock_guard<std::mutex> g(res_mu);
            res->ms_read_io += wall() - t0;
        }
        TR("window resolved");
        check_cancel(job);

        // ---- H2D + decompress (pipelined per sstable) ----
        hipStream_t copy_stream;
        HIP_CHECK(hipStreamCreate(&copy_stream));
        hipEvent_t e0, e1, e2, e3, e4;
        for (hipEvent_t* e : {&e0, &e1, &e2, &e3, &e4}) HIP_CHECK(hipEventCreate(e));
        HIP_CHECK(hipEventRecord(e0, stream));
        std::vector<DevBuf> d_comp(k), d_data(k), d_pos(k), d_chunks_s(k);
        DevBuf d_error;
        d_error.alloc(8);
        HIP_CHECK(hipMemsetAsync(d_error.p, 0, 8, stream));
        std::vector<hipEvent_t> ev_h2d(k);
        std::vector<std::vector<ChunkDesc>> chunks_all(k);  // kept alive past async copies
        double ms_read_data = 0;
        std::vector<const uint8_t*> vbase(k, nullptr);  // virtual decompressed origin
        for (int s = 0; s < k; s++) {
            double tr = wall();
            auto& ci = cinfos[s];
            uint64_t n_chunks_file = ci.offsets.size();
            uint64_t c_lo = win_chunk_lo[s];
            uint64_t c_hi = win_n[s] ? (data_hi[s] + CHUNK_LEN - 1) / CHUNK_LEN : c_lo;
            if (c_hi > n_chunks_file) c_hi = n_chunks_file;
            uint64_t n_wchunks = c_hi - c_lo;
            d_comp[s].alloc(comp_sz[s] ? comp_sz[s] : 1);
            if (adopt && win_comp_lo[s] == 0 && comp_sz[s] == su.comp_file_sz[s]) {
                // stripe-progressive H2D: copy each stripe as its read lands
                // (whole-file window only; stripe offsets are file-absolute)
                auto& sr = su.full_stripes[s];
                for (size_t t = 0; t < sr.th.size(); t++) {
                    if (sr.th[t].valid()) sr.th[t].wait();
                    if (sr.len[t])
                        HIP_CHECK(hipMemcpyAsync(d_comp[s].as<uint8_t>() + sr.off[t],
                                                 comp_pin[s] + sr.off[t], sr.len[t],
                                                 hipMemcpyHostToDevice, copy_stream));
                }
            } else if (adopt) {
                // token-restricted window over a preread file: wait for all
                // stripes, then one window-sized copy
                auto& sr = su.full_stripes[s];
                for (auto& t : sr.th)
                    if (t.valid()) t.wait();
                if (comp_sz[s])
                    HIP_CHECK(hipMemcpyAsync(d_comp[s].p, comp_pin[s], comp_sz[s],
                                             hipMemcpyHostToDevice, copy_stream));
            } else {
                data_readers[s].join();  // in-order wait; all reads run concurrently
                if (comp_sz[s])
                    HIP_CHECK(hipMemcpyAsync(d_comp[s].p, comp_pin[s], comp_sz[s],
                                             hipMemcpyHostToDevice, copy_stream));
            }
            ms_read_data = wall() - tr + ms_read_data;
            d_data[s].alloc(n_wchunks * (uint64_t)CHUNK_LEN + 16);
            // positions stay ABSOLUTE: the decompressed window is addressed
            // through a virtual origin so partition offsets need no rewrite
            vbase[s] = d_data[s].as<uint8_t>() - c_lo * (uint64_t)CHUNK_LEN;
            uint64_t n_wpos = (uint64_t)win_n[s] + 1;
            d_pos[s].alloc(n_wpos * 8);
            HIP_CHECK(hipMemcpyAsync(d_pos[s].p, positions[s].data() + win_blo[s], n_wpos * 8,
                                     hipMemcpyHostToDevice, copy_stream));
            auto& chunks = chunks_all[s];
            chunks.reserve(n_wchunks);
            for (uint64_t c = c_lo; c < c_hi; c++) {
                uint64_t off = ci.offsets[c];
                uint64_t end = c + 1 < n_chunks_file ? ci.offsets[c + 1] : su.comp_file_sz[s];
                ChunkDesc cd;
                cd.comp = d_comp[s].as<uint8_t>() + (off - win_comp_lo[s]);
                cd.out = d_data[s].as<uint8_t>() + (c - c_lo) * (uint64_t)CHUNK_LEN;
                cd.comp_len = (uint32_t)(end - off - 4);
                cd.out_len = (uint32_t)std::min<uint64_t>(CHUNK_LEN, ci.data_len - c * (uint64_t)CHUNK_LEN);
                chunks.push_back(cd);
            }
            d_chunks_s[s].alloc(chunks.size() * sizeof(ChunkDesc) + 16);
            if (!chunks.empty())
                HIP_CHECK(hipMemcpyAsync(d_chunks_s[s].p, chunks.data(), chunks.size() * sizeof(ChunkDesc),
                                         hipMemcpyHostToDevice, copy_stream));
            HIP_CHECK(hipEventCreate(&ev_h2d[s]));
            HIP_CHECK(hipEventRecord(ev_h2d[s], copy_stream));
            HIP_CHECK(hipStreamWaitEvent(stream, ev_h2d[s], 0));
            if (!chunks.empty()) {
                if (ci.snappy)
                    hipLaunchKernelGGL(k_snappy_decompress_chunks,
                                       dim3(lz4_decomp_grid((uint32_t)chunks.size(), 1)), dim3(WAVE), 0,
                                       stream, d_chunks_s[s].as<ChunkDesc>(), (uint32_t)chunks.size(), 1,
                                       d_error.as<unsigned long long>(), (const uint32_t*)g_crc256,
                                       (uint8_t*)nullptr);
                else
                    hipLaunchKernelGGL(k_lz4_decompress_wave, dim3(lz4_decomp_grid((uint32_t)chunks.size(), 1)), dim3(WAVE), 0,
                                       stream, d_chunks_s[s].as<ChunkDesc>(), (uint32_t)chunks.size(), 1,
                                       d_error.as<unsigned long long>(), (const uint32_t*)g_crc256);
            }
        }
        {
            std::lock_guard<std::mutex> g(res_mu);
            res->ms_read_io += ms_read_data;  // residual read wait not hidden by the pipeline
        }
        TR("ingest issued");
        HIP_CHECK(hipEventRecord(e1, stream));  // e0..e1: H2D+decompress pipeline (overlapped)
        HIP_CHECK(hipEventRecord(e2, stream));

        // corrupt inputs must fail cleanly, not fault the GPU: the parse
        // walk trusts decompressed bytes, so the chunk-CRC verdict is checked
        // BEFORE any parse kernel touches them
        {
            unsigned long long err = 0;
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(&err, d_error.p, 8, hipMemcpyDeviceToHost));
            if (err)
                throw std::runtime_error("input chunk decompress/CRC failed, code " +
                                         std::to_string(err));
        }

        // ---- parse (pass A: count + partition meta; pass B: row decode) ----
        uint64_t total_parts = 0;
        std::vector<SrcDesc2> srcs(k);
        for (int s = 0; s < k; s++) {
            srcs[s].data = vbase[s];
            srcs[s].part_pos = d_pos[s].as<uint64_t>();
            srcs[s].n_parts = win_n[s];
            srcs[s].min_ts = stats[s].hdr_min_ts;
            srcs[s].min_ldt = stats[s].hdr_min_ldt;
            srcs[s].min_ttl = stats[s].hdr_min_ttl;
            srcs[s].rec_base = (uint32_t)total_parts;
            total_parts += srcs[s].n_parts;
        }
        if (total_parts > 0xFFFFFFFFull) throw std::runtime_error("too many partitions for one job");
        SchemaParams sch{};
        sch.n_ck = (uint32_t)su.ck_widths.size();
        DevBuf d_ck_w;
        d_ck_w.alloc(su.ck_widths.size() * 4 + 8);
        if (sch.n_ck)
            HIP_CHECK(hipMemcpyAsync(d_ck_w.p, su.ck_widths.data(), su.ck_widths.size() * 4,
                                     hipMemcpyHostToDevice, stream));
        sch.ck_w = d_ck_w.as<int32_t>();
        sch.n_cols = (uint32_t)col_fixed_h.size();
        sch.n_cpx = su.n_cpx;
        sch.counters = su.counters ? 1u : 0u;
        DevBuf d_col_fixed;
        d_col_fixed.alloc(col_fixed_h.size() * 4);
        HIP_CHECK(hipMemcpyAsync(d_col_fixed.p, col_fixed_h.data(), col_fixed_h.size() * 4,
                                 hipMemcpyHostToDevice, stream));
        sch.col_fixed = d_col_fixed.as<int32_t>();
        sch.n_static = (uint32_t)su.static_fixed_h.size();
        DevBuf d_static_fixed;
        d_static_fixed.alloc(su.static_fixed_h.size() * 4 + 8);
        if (sch.n_static)
            HIP_CHECK(hipMemcpyAsync(d_static_fixed.p, su.static_fixed_h.data(),
                                     su.static_fixed_h.size() * 4, hipMemcpyHostToDevice, stream));
        sch.static_fixed = d_static_fixed.as<int32_t>();
        sch.column_index_size = 64 * 1024;
        DevBuf d_srcs, d_recs_a, d_recs_b, d_rows_in;
        d_srcs.alloc(srcs.size() * sizeof(SrcDesc2));
        HIP_CHECK(hipMemcpyAsync(d_srcs.p, srcs.data(), srcs.size() * sizeof(SrcDesc2),
                                 hipMemcpyHostToDevice, stream));
        d_recs_a.alloc(total_parts * sizeof(MRec));
        d_recs_b.alloc(total_parts * sizeof(MRec));
        d_rows_in.alloc(8);
        HIP_CHECK(hipMemsetAsync(d_rows_in.p, 0, 8, stream));
        DevBuf p_pdm, p_pdl, p_rcnt, p_rbase;
        ParsedCols pc{};
        p_pdm.alloc(total_parts * 8); pc.pdel_mfda = p_pdm.as<int64_t>();
        p_pdl.alloc(total_parts * 4); pc.pdel_ldt = p_pdl.as<uint32_t>();
        p_rcnt.alloc(total_parts * 4); pc.row_count = p_rcnt.as<uint32_t>();
        p_rbase.alloc(total_parts * 8); pc.row_base = p_rbase.as<uint64_t>();
        DevBuf p_kaddr;
        p_kaddr.alloc(total_parts * 8); pc.key_addr = p_kaddr.as<uint64_t>();
        StaticColsBuf p_static;
        p_static.alloc(sch.n_static ? total_parts : 1, sch.n_static);
        pc.st = p_static.st;
        {
            uint32_t blocks = (uint32_t)((total_parts + 255) / 256);
            hipLaunchKernelGGL(k_parse_count, dim3(blocks), dim3(256), 0, stream,
                               d_srcs.as<SrcDesc2>(), (uint32_t)k, (uint32_t)total_parts,
                               d_recs_a.as<MRec>(), pc, sch, d_error.as<unsigned long long>());
            hipLaunchKernelGGL(k_widen_u32, dim3(blocks), dim3(256), 0, stream,
                               pc.row_count, pc.row_base, total_parts);
        }
        uint64_t total_in_rows = exscan_u64(pc.row_base, total_parts, stream);
        UnfColsBuf in_rows;
        in_rows.alloc(total_in_rows, sch.n_cols, sch.n_ck, sch.n_cpx);
        DevBuf p_cpxtot, p_cpxbase;
        uint64_t total_in_cpx = 0;
        {
            uint32_t blocks = (uint32_t)((total_parts + 255) / 256);
            if (sch.n_cpx) {
                // pass B1 (COUNT): full walk, per-partition complex-cell totals
                p_cpxtot.alloc(total_parts * 4 + 8);
                pc.cpx_total = p_cpxtot.as<uint32_t>();
                hipLaunchKernelGGL(k_parse_rows, dim3(blocks), dim3(256), 0, stream,
                                   d_srcs.as<SrcDesc2>(), (uint32_t)k, (uint32_t)total_parts, pc,
                                   in_rows.uc, sch, d_error.as<unsigned long long>(),
                                   d_rows_in.as<unsigned long long>(), nullptr);
                HIP_CHECK(hipMemsetAsync(d_rows_in.p, 0, 8, stream));  // recounted by pass B2
                p_cpxbase.alloc(total_parts * 8 + 8);
                hipLaunchKernelGGL(k_widen_u32, dim3(blocks), dim3(256), 0, stream,
                                   pc.cpx_total, p_cpxbase.as<uint64_t>(), total_parts);
                total_in_cpx = exscan_u64(p_cpxbase.as<uint64_t>(), total_parts, stream);
                in_rows.alloc_cpx_arena(total_in_cpx);
            }
            hipLaunchKernelGGL(k_parse_rows, dim3(blocks), dim3(256), 0, stream,
                               d_srcs.as<SrcDesc2>(), (uint32_t)k, (uint32_t)total_parts, pc,
                               in_rows.uc, sch, d_error.as<unsigned long long>(),
                               d_rows_in.as<unsigned long long>(),
                               sch.n_cpx ? p_cpxbase.as<uint64_t>() : nullptr);
        }
        HIP_CHECK(hipEventRecord(e3, stream));
        {
            unsigned long long err = 0;
            HIP_CHECK(hipStreamSynchronize(stream));
            TR("parse synced");
        check_cancel(job);
            HIP_CHECK(hipMemcpy(&err, d_error.p, 8, hipMemcpyDeviceToHost));
            if (err) throw std::runtime_error("GPU decode/parse error code " + std::to_string(err));
            uint64_t rows_in_local = 0;
            HIP_CHECK(hipMemcpy(&rows_in_local, d_rows_in.p, 8, hipMemcpyDeviceToHost));
            std::lock_guard<std::mutex> g(res_mu);
            res->rows_in += rows_in_local;
        }

        // exact-key comparator fallback: point the merge kernels at the
        // decompressed inputs (keys > 8 bytes tie-break by byte walk)
        KeyLut lut{};
        for (int s2 = 0; s2 < k; s2++) {
            lut.base[s2] = vbase[s2];
            lut.pos[s2] = d_pos[s2].as<uint64_t>();
        }
        lut.enabled = 1;

        // ---- merge (pairwise rounds over pre-sorted source runs) ----
        // In garbage-collect mode the first k_data inputs are the compacted
        // data; the rest are tombstone sources merged in a second pass.
        int kd = su.k_data > 0 ? su.k_data : k;
        bool gc_mode = kd < k;
        uint64_t data_parts = 0;
        for (int s = 0; s < kd; s++) data_parts += srcs[s].n_parts;
        std::vector<uint64_t> runs;
        runs.push_back(0);
        for (int s = 0; s < kd; s++) runs.push_back(runs.back() + srcs[s].n_parts);
        MRec* d_sorted = merge_sorted_runs(d_recs_a.as<MRec>(), d_recs_b.as<MRec>(), runs, stream, lut);
        TR("merge issued");

        // ---- group heads + starts ----
        DevBuf d_head, d_gstart, d_ngroups;
        d_head.alloc(data_parts * 8 + 8);
        d_gstart.alloc(data_parts * 8 + 8);
        d_ngroups.alloc(8);
        {
            uint32_t blocks = (uint32_t)((data_parts + 255) / 256);
            hipLaunchKernelGGL(k_group_heads, dim3(blocks), dim3(256), 0, stream, d_sorted,
                               data_parts, d_head.as<uint64_t>(), lut);
        }
        exscan_u64(d_head.as<uint64_t>(), data_parts, stream);
        {
            uint32_t blocks = (uint32_t)((data_parts + 255) / 256);
            hipLaunchKernelGGL(k_group_starts2, dim3(blocks), dim3(256), 0, stream, d_sorted,
                               data_parts, d_head.as<uint64_t>(), d_gstart.as<uint64_t>(),
                               d_ngroups.as<uint64_t>(), lut);
        }
        uint64_t n_groups = 0;
        HIP_CHECK(hipStreamSynchronize(stream));
        HIP_CHECK(hipMemcpy(&n_groups, d_ngroups.p, 8, hipMemcpyDeviceToHost));
        TR("groups known");
        check_cancel(job);

        // ---- reconcile + purge ----
        // Host staging for the small async H2D setup copies below. These must
        // outlive the copies until the next hipStreamSynchronize on `stream`
        // (an async copy from freed pageable memory is undefined), so they
        // live at this scope, past the reconcile-phase sync.
        std::vector<uint32_t> bases(k);
        std::vector<int64_t> klo_h, khi_h, ov_lo_h, ov_hi_h, ov_ts_h;
        DevBuf d_srcbases, d_group_rows;
        {
            for (int s = 0; s < k; s++) bases[s] = srcs[s].rec_base;
            d_srcbases.alloc(k * 4);
            HIP_CHECK(hipMemcpyAsync(d_srcbases.p, bases.data(), k * 4, hipMemcpyHostToDevice, stream));
        }
        d_group_rows.alloc(n_groups * 8);
        {
            uint32_t blocks = (uint32_t)((n_groups + 255) / 256);
            hipLaunchKernelGGL(k_group_row_sums, dim3(blocks), dim3(256), 0, stream, d_sorted,
                               d_gstart.as<uint64_t>(), n_groups, data_parts,
                               d_srcbases.as<uint32_t>(), pc, d_group_rows.as<uint64_t>());
        }
        uint64_t total_out_rows = exscan_u64(d_group_rows.as<uint64_t>(), n_groups, stream);
        TR("out rows scanned");
        OutPartsBuf opb;
        opb.alloc(n_groups, sch.n_static);
        UnfColsBuf out_rows;
        out_rows.alloc(total_out_rows, sch.n_cols, sch.n_ck, sch.n_cpx);
        if (sch.n_cpx) out_rows.alloc_cpx_arena(total_in_cpx);
        DevBuf d_stats, d_tomb;
        d_stats.alloc(sizeof(OutStats));
        init_outstats(d_stats, stream);
        DevBuf d_ctr_arena;
        uint64_t ctr_arena_cap = 0;
        if (sch.counters) {
            // merged-context arena: capacity = total input counter bytes
            // (any k-way merge result is bounded by the sum of its inputs)
            DevBuf d_sum;
            d_sum.alloc(8);
            HIP_CHECK(hipMemsetAsync(d_sum.p, 0, 8, stream));
            uint64_t n_cells = total_in_rows * sch.n_cols;
            if (n_cells) {
                uint32_t blocks2 = (uint32_t)((n_cells + 255) / 256);
                hipLaunchKernelGGL(k_sum_vallen, dim3(blocks2), dim3(256), 0, stream,
                                   in_rows.uc, n_cells, d_sum.as<unsigned long long>());
            }
            unsigned long long total_ctr = 0;
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(&total_ctr, d_sum.p, 8, hipMemcpyDeviceToHost));
            ctr_arena_cap = total_ctr + 64;
            d_ctr_arena.alloc(ctr_arena_cap);
            sch.ctr_arena = d_ctr_arena.as<uint8_t>();
        }
        uint32_t tomb_cap = (uint32_t)std::min<uint64_t>(
            total_out_rows * ((uint64_t)sch.n_cols + sch.n_cpx + 2) + total_in_cpx +
                n_groups * ((uint64_t)sch.n_static + 2) + 1024,
            400000000ull);
        d_tomb.alloc((uint64_t)tomb_cap * 4);
        DevBuf d_ov_lo, d_ov_hi, d_ov_ts;
        PurgeParams2 pp{};
        pp.now_sec = job->now_sec;
        pp.gc_before = job->gc_before;
        pp.never_purge = job->never_purge;
        pp.enforce_strict_liveness = job->enforce_strict_liveness;
        pp.n_overlaps = job->n_overlaps;
        pp.has_shard = job->has_token_range;
        pp.shard_lo = job->token_lo;
        pp.shard_hi = job->token_hi;
        if (shard_exact) {
            // da internal shards: trie-separator windows are one partition
            // wide of exact; this token filter (on GPU-computed tokens) is
            // the exact boundary, intersected with any job-level range
            if (pp.has_shard) {
                pp.shard_lo = std::max(pp.shard_lo, sh_lo);
                pp.shard_hi = std::min(pp.shard_hi, sh_hi);
            } else {
                pp.has_shard = 1;
                pp.shard_lo = sh_lo;
                pp.shard_hi = sh_hi;
            }
        }
        DevBuf d_kr_lo, d_kr_hi;
        if (job->n_keep_ranges > 0) {
            klo_h.resize(job->n_keep_ranges);
            khi_h.resize(job->n_keep_ranges);
            for (int i = 0; i < job->n_keep_ranges; i++) {
                klo_h[i] = job->keep_ranges[i].token_lo;
                khi_h[i] = job->keep_ranges[i].token_hi;
            }
            d_kr_lo.alloc(job->n_keep_ranges * 8);
            d_kr_hi.alloc(job->n_keep_ranges * 8);
            HIP_CHECK(hipMemcpyAsync(d_kr_lo.p, klo_h.data(), job->n_keep_ranges * 8, hipMemcpyHostToDevice, stream));
            HIP_CHECK(hipMemcpyAsync(d_kr_hi.p, khi_h.data(), job->n_keep_ranges * 8, hipMemcpyHostToDevice, stream));
            pp.kr_lo = d_kr_lo.as<int64_t>();
            pp.kr_hi = d_kr_hi.as<int64_t>();
            pp.n_keep_ranges = job->n_keep_ranges;
            pp.invert_ranges = job->invert_ranges;
        }
        DevBuf d_ov_bw, d_ov_boff, d_ov_bbits, d_ov_bk;
        std::vector<uint32_t> ov_bw_h;
        std::vector<uint64_t> ov_boff_h, ov_bbits_h;
        std::vector<int32_t> ov_bk_h;
        if (job->n_overlaps > 0) {
            ov_lo_h.resize(job->n_overlaps);
            ov_hi_h.resize(job->n_overlaps);
            ov_ts_h.resize(job->n_overlaps);
            bool any_bloom = false;
            for (int i = 0; i < job->n_overlaps; i++) {
                ov_lo_h[i] = job->overlaps[i].token_lo;
                ov_hi_h[i] = job->overlaps[i].token_hi;
                ov_ts_h[i] = job->overlaps[i].min_timestamp;
                any_bloom |= job->overlaps[i].bloom_bits != nullptr;
            }
            d_ov_lo.alloc(job->n_overlaps * 8);
            d_ov_hi.alloc(job->n_overlaps * 8);
            d_ov_ts.alloc(job->n_overlaps * 8);
            HIP_CHECK(hipMemcpyAsync(d_ov_lo.p, ov_lo_h.data(), job->n_overlaps * 8, hipMemcpyHostToDevice, stream));
            HIP_CHECK(hipMemcpyAsync(d_ov_hi.p, ov_hi_h.data(), job->n_overlaps * 8, hipMemcpyHostToDevice, stream));
            HIP_CHECK(hipMemcpyAsync(d_ov_ts.p, ov_ts_h.data(), job->n_overlaps * 8, hipMemcpyHostToDevice, stream));
            pp.ov_lo = d_ov_lo.as<int64_t>();
            pp.ov_hi = d_ov_hi.as<int64_t>();
            pp.ov_min_ts = d_ov_ts.as<int64_t>();
            if (any_bloom) {
                // pack the per-overlap bloom bit arrays into one device
                // buffer (word offsets per entry; bit_len 0 = no bloom)
                for (int i = 0; i < job->n_overlaps; i++) {
                    const auto& ov = job->overlaps[i];
                    ov_boff_h.push_back(ov_bw_h.size());
                    if (ov.bloom_bits && ov.bloom_bit_len) {
                        uint64_t words32 = (ov.bloom_bit_len + 31) / 32;
                        ov_bw_h.insert(ov_bw_h.end(), ov.bloom_bits, ov.bloom_bits + words32);
                        ov_bbits_h.push_back(ov.bloom_bit_len);
                        ov_bk_h.push_back(ov.bloom_hash_count);
                    } else {
                        ov_bbits_h.push_back(0);
                        ov_bk_h.push_back(0);
                    }
                }
                d_ov_bw.alloc(ov_bw_h.size() * 4 + 8);
                d_ov_boff.alloc(ov_boff_h.size() * 8);
                d_ov_bbits.alloc(ov_bbits_h.size() * 8);
                d_ov_bk.alloc(ov_bk_h.size() * 4);
                HIP_CHECK(hipMemcpyAsync(d_ov_bw.p, ov_bw_h.data(), ov_bw_h.size() * 4, hipMemcpyHostToDevice, stream));
                HIP_CHECK(hipMemcpyAsync(d_ov_boff.p, ov_boff_h.data(), ov_boff_h.size() * 8, hipMemcpyHostToDevice, stream));
                HIP_CHECK(hipMemcpyAsync(d_ov_bbits.p, ov_bbits_h.data(), ov_bbits_h.size() * 8, hipMemcpyHostToDevice, stream));
                HIP_CHECK(hipMemcpyAsync(d_ov_bk.p, ov_bk_h.data(), ov_bk_h.size() * 4, hipMemcpyHostToDevice, stream));
                pp.ov_bloom_words = d_ov_bw.as<uint32_t>();
                pp.ov_bloom_off = d_ov_boff.as<uint64_t>();
                pp.ov_bloom_bits = d_ov_bbits.as<uint64_t>();
                pp.ov_bloom_k = d_ov_bk.as<int32_t>();
                pp.ov_has_bloom = 1;
            }
        }
        hipEvent_t er0, er1;
        HIP_CHECK(hipEventCreate(&er0));
        HIP_CHECK(hipEventCreate(&er1));
        HIP_CHECK(hipEventRecord(er0, stream));
        // GC mode: purge is DEFERRED until after the garbage filter
        // (GarbageSkipper runs before the Purger in CompactionIterator)
        PurgeParams2 pp_data = pp;
        if (gc_mode) {
            pp_data.gc_before = INT64_MIN;
            pp_data.never_purge = 1;
            pp_data.enforce_strict_liveness = 0;
        }
        {
            uint32_t blocks = (uint32_t)((n_groups + 255) / 256);
            auto launch_rec = [&](auto kern) {
                hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), 0, stream, d_sorted,
                                   d_gstart.as<uint64_t>(), n_groups, data_parts,
                                   d_srcbases.as<uint32_t>(), pc, in_rows.uc, opb.op, out_rows.uc,
                                   d_group_rows.as<uint64_t>(), sch, pp_data, d_stats.as<OutStats>(),
                                   d_error.as<unsigned long long>());
            };
            if (kd <= 8) launch_rec(k_reconcile_rows<8>);
            else if (kd <= 16) launch_rec(k_reconcile_rows<16>);
            else launch_rec(k_reconcile_rows<64>);
        }
        HIP_CHECK(hipEventRecord(er1, stream));
        HIP_CHECK(hipEventRecord(e4, stream));
        TR("reconcile issued");
        HIP_CHECK(hipStreamSynchronize(stream));
        {
            unsigned long long err = 0;
            HIP_CHECK(hipMemcpy(&err, d_error.p, 8, hipMemcpyDeviceToHost));
            if (err) throw std::runtime_error("GPU reconcile error code " + std::to_string(err));
        }

        // ---- garbage collect: merge the tombstone sources, filter, purge ----
        UnfColsBuf gc_rows;          // filtered data arena (replaces out_rows)
        UnfColsBuf* rows_for_writer = &out_rows;
        OutPartsBuf opb_t;
        UnfColsBuf out_rows_t;
        DevBuf d_ctr_arena_t;
        if (gc_mode && n_groups > 0) {
            uint64_t src_parts = total_parts - data_parts;
            uint64_t n_groups_t = 0;
            if (src_parts > 0) {
                std::vector<uint64_t> runs_t;
                runs_t.push_back(0);
                for (int s2 = kd; s2 < k; s2++) runs_t.push_back(runs_t.back() + srcs[s2].n_parts);
                MRec* d_sorted_t = merge_sorted_runs(d_recs_a.as<MRec>() + data_parts,
                                                     d_recs_b.as<MRec>() + data_parts, runs_t,
                                                     stream, lut);
                DevBuf d_head_t, d_gstart_t, d_ng_t;
                d_head_t.alloc(src_parts * 8 + 8);
                d_gstart_t.alloc(src_parts * 8 + 8);
                d_ng_t.alloc(8);
                uint32_t blocks_t = (uint32_t)((src_parts + 255) / 256);
                hipLaunchKernelGGL(k_group_heads, dim3(blocks_t), dim3(256), 0, stream, d_sorted_t,
                                   src_parts, d_head_t.as<uint64_t>(), lut);
                exscan_u64(d_head_t.as<uint64_t>(), src_parts, stream);
                hipLaunchKernelGGL(k_group_starts2, dim3(blocks_t), dim3(256), 0, stream, d_sorted_t,
                                   src_parts, d_head_t.as<uint64_t>(), d_gstart_t.as<uint64_t>(),
                                   d_ng_t.as<uint64_t>(), lut);
                HIP_CHECK(hipStreamSynchronize(stream));
                HIP_CHECK(hipMemcpy(&n_groups_t, d_ng_t.p, 8, hipMemcpyDeviceToHost));
                DevBuf d_grows_t, d_stats_t;
                d_grows_t.alloc(n_groups_t * 8 + 8);
                {
                    uint32_t blocks_g = (uint32_t)((n_groups_t + 255) / 256);
                    hipLaunchKernelGGL(k_group_row_sums, dim3(blocks_g), dim3(256), 0, stream,
                                       d_sorted_t, d_gstart_t.as<uint64_t>(), n_groups_t, src_parts,
                                       d_srcbases.as<uint32_t>(), pc, d_grows_t.as<uint64_t>());
                }
                uint64_t t_out_rows = exscan_u64(d_grows_t.as<uint64_t>(), n_groups_t, stream);
                opb_t.alloc(n_groups_t ? n_groups_t : 1, sch.n_static);
                out_rows_t.alloc(t_out_rows, sch.n_cols, sch.n_ck, sch.n_cpx);
                if (sch.n_cpx) out_rows_t.alloc_cpx_arena(total_in_cpx);
                d_stats_t.alloc(sizeof(OutStats));
                init_outstats(d_stats_t, stream);
                // the source pass bump-allocates merged counter contexts from
                // its own arena (the main pass's bump counter lives in the
                // OTHER OutStats, so sharing one arena would overlap)
                SchemaParams sch_t = sch;
                if (sch.counters) {
                    d_ctr_arena_t.alloc(ctr_arena_cap);
                    sch_t.ctr_arena = d_ctr_arena_t.as<uint8_t>();
                }
                PurgeParams2 pp_src = pp_data;
                pp_src.has_shard = 0;  // sources shadow regardless of the shard
                {
                    uint32_t blocks_g = (uint32_t)((n_groups_t + 255) / 256);
                    auto launch_rec_t = [&](auto kern) {
                        hipLaunchKernelGGL(kern, dim3(blocks_g), dim3(256), 0, stream, d_sorted_t,
                                           d_gstart_t.as<uint64_t>(), n_groups_t, src_parts,
                                           d_srcbases.as<uint32_t>(), pc, in_rows.uc, opb_t.op,
                                           out_rows_t.uc, d_grows_t.as<uint64_t>(), sch_t, pp_src,
                                           d_stats_t.as<OutStats>(),
                                           d_error.as<unsigned long long>());
                    };
                    int ks = k - kd;
                    if (ks <= 8) launch_rec_t(k_reconcile_rows<8>);
                    else if (ks <= 16) launch_rec_t(k_reconcile_rows<16>);
                    else launch_rec_t(k_reconcile_rows<64>);
                }
            }
            // match + capacity scan + filter
            DevBuf d_tidx, d_cap;
            d_tidx.alloc(n_groups * 8);
            d_cap.alloc(n_groups * 8);
            {
                uint32_t blocks_g = (uint32_t)((n_groups + 255) / 256);
                hipLaunchKernelGGL(k_gc_match, dim3(blocks_g), dim3(256), 0, stream, opb.op,
                                   n_groups, opb_t.op, n_groups_t, d_tidx.as<int64_t>(),
                                   d_cap.as<uint64_t>());
            }
            uint64_t gc_total = exscan_u64(d_cap.as<uint64_t>(), n_groups, stream);
            gc_rows.alloc(gc_total, sch.n_cols, sch.n_ck, sch.n_cpx);
            // the cpx CELL arena is shared: the filter compacts each row's
            // own [start,count) segment of out_rows' arena in place
            if (sch.n_cpx) gc_rows.uc.cpx = out_rows.uc.cpx;
            {
                uint32_t blocks_g = (uint32_t)((n_groups + 255) / 256);
                hipLaunchKernelGGL(k_garbage_filter, dim3(blocks_g), dim3(256), 0, stream, opb.op,
                                   n_groups, opb_t.op, out_rows.uc, out_rows_t.uc, gc_rows.uc,
                                   d_tidx.as<int64_t>(), d_cap.as<uint64_t>(), sch,
                                   job->cell_level_gc ? 1 : 0);
            }
            // reset first/last group, then the deferred purge pass
            HIP_CHECK(hipMemsetAsync((uint8_t*)d_stats.p + offsetof(OutStats, first_group), 0xFF, 8, stream));
            HIP_CHECK(hipMemsetAsync((uint8_t*)d_stats.p + offsetof(OutStats, last_group), 0x00, 8, stream));
            {
                uint32_t blocks_g = (uint32_t)((n_groups + 255) / 256);
                hipLaunchKernelGGL(k_purge_parts, dim3(blocks_g), dim3(256), 0, stream, opb.op,
                                   gc_rows.uc, n_groups, sch, pp, d_stats.as<OutStats>());
            }
            HIP_CHECK(hipStreamSynchronize(stream));
            {
                unsigned long long err = 0;
                HIP_CHECK(hipMemcpy(&err, d_error.p, 8, hipMemcpyDeviceToHost));
                if (err) throw std::runtime_error("GPU gc filter error code " + std::to_string(err));
            }
            rows_for_writer = &gc_rows;
            TR("garbage filtered");
        }

        // ---- output header (SerializationHeader.make: desc-generation stats merge) ----
        SerParams2 sp2{};
        sp2.sch = sch;
        {
            std::vector<int> order(kd);
            for (int s = 0; s < kd; s++) order[s] = s;
            std::stable_sort(order.begin(), order.end(),
                             [&](int a, int b) { return generations[a] > generations[b]; });
            int64_t min_ts = INT64_MAX, min_ldt = INT64_MAX;
            int32_t min_ttl = INT32_MAX;
            for (int s : order) {
                min_ts = std::min(min_ts, stats[s].min_timestamp);
                min_ldt = std::min(min_ldt, stats[s].min_ldt);
                min_ttl = std::min(min_ttl, stats[s].min_ttl);
            }
            sp2.hs.min_ts = min_ts == NO_TIMESTAMP ? TIMESTAMP_EPOCH : min_ts;
            sp2.hs.min_ldt = min_ldt == NO_DELETION_TIME ? DELETION_TIME_EPOCH : min_ldt;
            sp2.hs.min_ttl = min_ttl == INT32_MAX ? 0 : min_ttl;
        }
        if (g_validate_out) {
            // VALIDATION epilogue: per-partition repair digests instead of a
            // written sstable (k_validate_digest; Validator.rowHash)
            bytes names;
            std::vector<uint32_t> noff;
            for (auto& [nm, t2] : stats[0].regular_cols) {
                (void)t2;
                noff.push_back((uint32_t)names.size());
                names.insert(names.end(), nm.begin(), nm.end());
            }
            for (auto& [nm, t2] : stats[0].static_cols) {
                (void)t2;
                noff.push_back((uint32_t)names.size());
                names.insert(names.end(), nm.begin(), nm.end());
            }
            noff.push_back((uint32_t)names.size());
            DevBuf d_names, d_noff, d_hash;
            d_names.alloc(names.size() + 8);
            d_noff.alloc(noff.size() * 4);
            d_hash.alloc(n_groups * 32 + 32);
            HIP_CHECK(hipMemcpyAsync(d_names.p, names.data(), names.size(), hipMemcpyHostToDevice, stream));
            HIP_CHECK(hipMemcpyAsync(d_noff.p, noff.data(), noff.size() * 4, hipMemcpyHostToDevice, stream));
            ValidateParams vpar{};
            vpar.names = d_names.as<uint8_t>();
            vpar.name_off = d_noff.as<uint32_t>();
            vpar.n_reg = (uint32_t)stats[0].regular_cols.size();
            vpar.n_static = (uint32_t)stats[0].static_cols.size();
            vpar.counters = sch.counters;
            vpar.n_cpx = sch.n_cpx;
            vpar.n_ck = sch.n_ck;
            vpar.ck_w = sch.ck_w;
            {
                uint32_t blocks2 = (uint32_t)((n_groups + 255) / 256);
                hipLaunchKernelGGL(k_validate_digest, dim3(blocks2), dim3(256), 0, stream, opb.op,
                                   rows_for_writer->uc, n_groups, vpar, d_hash.as<uint8_t>());
            }
            std::vector<uint8_t> h_hash(n_groups * 32);
            std::vector<int64_t> h_tok(n_groups);
            std::vector<uint8_t> h_keep(n_groups);
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(h_hash.data(), d_hash.p, n_groups * 32, hipMemcpyDeviceToHost));
            HIP_CHECK(hipMemcpy(h_tok.data(), opb.op.token, n_groups * 8, hipMemcpyDeviceToHost));
            HIP_CHECK(hipMemcpy(h_keep.data(), opb.op.keep, n_groups, hipMemcpyDeviceToHost));
            bytes outb;
            uint64_t cnt = 0;
            for (uint64_t g2 = 0; g2 < n_groups; g2++) {
                if (!h_keep[g2]) continue;
                for (int b2 = 7; b2 >= 0; b2--) outb.push_back((uint8_t)((uint64_t)h_tok[g2] >> (8 * b2)));
                outb.insert(outb.end(), h_hash.begin() + g2 * 32, h_hash.begin() + g2 * 32 + 32);
                cnt++;
            }
            write_file(g_validate_out, outb.data(), outb.size());
            g_validate_count = cnt;
            {
                std::lock_guard<std::mutex> gl(res_mu);
                res->partitions_out += cnt;
            }
            HIP_CHECK(hipStreamDestroy(stream));
            return;
        }
        WriteDeviceOut w = write_sstable_device(opb, *rows_for_writer, n_groups, sp2, d_stats, d_tomb,
                                                tomb_cap, out_base_str, stats[0].key_type,
                                                stats[0].clustering_types, stats[0].regular_cols,
                                                stats[0].static_cols, stream, wslot,
                                                su.cinfos[0].snappy, su.bti);
        TR("writer done");
        {
            OutStats hst;
            HIP_CHECK(hipMemcpy(&hst, d_stats.p, sizeof(OutStats), hipMemcpyDeviceToHost));
            float th2d, tdec, tparse, tmerge, trec;
            HIP_CHECK(hipEventElapsedTime(&th2d, e0, e1));
            HIP_CHECK(hipEventElapsedTime(&tdec, e1, e2));
            HIP_CHECK(hipEventElapsedTime(&tparse, e2, e3));
            HIP_CHECK(hipEventElapsedTime(&tmerge, e3, e4));
            HIP_CHECK(hipEventElapsedTime(&trec, er0, er1));
            std::lock_guard<std::mutex> g(res_mu);
            res->partitions_out += w.partitions;
            res->rows_out += w.rows;
            res->output_uncompressed_bytes += w.uncompressed_len;
            res->output_compressed_bytes += w.compressed_len;
            for (int i = 0; i < 64; i++) res->merged_counts[i] += hst.merged_counts[i];
            res->ms_h2d = 0;         // overlapped into the ingest pipeline (e0..e1)
            res->ms_decompress += th2d + tdec;  // read/H2D/decompress pipeline, GPU side
            res->ms_parse += tparse;
            res->ms_reconcile += trec;
            res->ms_merge += tmerge - trec;  // merge rounds + grouping
            res->ms_serialize += w.ms_sizes + w.ms_serialize;
            res->ms_compress += w.ms_compress;
            res->ms_d2h += w.ms_d2h;
            res->ms_write_io += w.ms_io;
            // dominant kernel estimate
            struct { const char* n; double ms; } ks[] = {
                {"k_lz4_decompress", (double)th2d + tdec}, {"k_parse", (double)tparse},
                {"merge+reconcile", (double)tmerge},
                {"k_serialize", w.ms_serialize}, {"k_lz4_compress", w.ms_compress}};
            for (auto& kk : ks)
                if (kk.ms > res->dominant_kernel_ms) {
                    res->dominant_kernel_ms = kk.ms;
                    snprintf(res->dominant_kernel, sizeof(res->dominant_kernel), "%s", kk.n);
                }
            res->dominant_kernel_launches = 1;
        }
        for (hipEvent_t e : {e0, e1, e2, e3, e4, er0, er1}) (void)hipEventDestroy(e);
        for (int s = 0; s < k; s++) (void)hipEventDestroy(ev_h2d[s]);
        HIP_CHECK(hipStreamDestroy(copy_stream));
    }
    HIP_CHECK(hipStreamDestroy(stream));
}



// ---------------------------------------------------------------------------
// compaction setup (host metadata shared by all shards) + sharded driver
// ---------------------------------------------------------------------------

static void compact_setup(const gpuc_job* job, CompactSetup& su, bool preread_full) {
    int k = job->n_inputs + (job->n_tomb_sources > 0 ? job->n_tomb_sources : 0);
    su.k = k;
    {
        auto exists = [](const std::string& p2) {
            FILE* f = fopen(p2.c_str(), "rb");
            if (f) fclose(f);
            return f != nullptr;
        };
        std::string b0 = job->input_bases[0];
        su.bti = exists(b0 + "-Partitions.db") && !exists(b0 + "-Index.db");
    }
    su.k_data = job->n_inputs;
    su.in_bases.resize(k);
    su.index_data.resize(k);
    su.cinfos.resize(k);
    su.stats.resize(k);
    su.generations.resize(k);
    su.positions.resize(k);
    su.entry_offs.resize(k);
    su.bti_prefixes.resize(k);
    su.comp_file_sz.resize(k);
    // Data.db reads start FIRST (they are the long pole); metadata parse and
    // index decode run while they stream into the pinned arenas
    if (preread_full) {
        su.full_pin.resize(k, nullptr);
        su.full_stripes.resize(k);
        struct RTask { int s; uint64_t off, len; std::promise<void> done; };
        auto tasks = std::make_shared<std::vector<RTask>>();
        for (int s = 0; s < k; s++) {
            su.in_bases[s] = s < su.k_data ? job->input_bases[s]
                                           : job->tombstone_source_bases[s - su.k_data];
            su.comp_file_sz[s] = file_size_of(su.in_bases[s] + "-Data.db");
            su.full_pin[s] = (uint8_t*)g_pin_in[0][s].get(su.comp_file_sz[s] ? su.comp_file_sz[s] : 1);
            if (!su.full_pin[s]) throw std::runtime_error("pinned alloc failed");
            auto& sr = su.full_stripes[s];
            const int NS_STRIPE = 6;
            uint64_t per = (su.comp_file_sz[s] + NS_STRIPE - 1) / NS_STRIPE;
            for (int t = 0; t < NS_STRIPE; t++) {
                uint64_t o = (uint64_t)t * per;
                if (o >= su.comp_file_sz[s]) break;
                uint64_t len = std::min<uint64_t>(per, su.comp_file_sz[s] - o);
                sr.off.push_back(o);
                sr.len.push_back(len);
                tasks->push_back(RTask{s, o, len, std::promise<void>()});
                sr.th.push_back(tasks->back().done.get_future());
            }
        }
        // pool drains tasks in queue order (file-major): early files complete
        // first and the consumer's per-stripe waits overlap the later reads
        auto next = std::make_shared<std::atomic<size_t>>(0);
        int nworkers = (int)std::min<size_t>(12, tasks->size());
        for (int wkr = 0; wkr < nworkers; wkr++) {
            su.read_pool.emplace_back([&su, tasks, next] {
                for (;;) {
                    size_t i = next->fetch_add(1);
                    if (i >= tasks->size()) return;
                    RTask& rt = (*tasks)[i];
                    read_file_range(su.in_bases[rt.s] + "-Data.db", su.full_pin[rt.s] + rt.off,
                                    rt.off, rt.len, 1);
                    rt.done.set_value();
                }
            });
        }
    }
    {
        std::vector<std::thread> mth;
        std::vector<std::string> merr(k);
        for (int s = 0; s < k; s++) {
            su.in_bases[s] = s < su.k_data ? job->input_bases[s]
                                           : job->tombstone_source_bases[s - su.k_data];
            mth.emplace_back([&, s, preread_full]() {
                try {
                    const std::string& base = su.in_bases[s];
                    if (su.bti) {
                        // `da` inputs: partition positions come from the
                        // Partitions.db trie (payload < 0 == ~data_pos;
                        // >= 0 == Rows.db TrieIndexEntry footer, whose
                        // dataStartPosition leads). DFS order == key order
                        // == data order.
                        bytes pf = read_file(base + "-Partitions.db");
                        bytes rf = read_file(base + "-Rows.db");
                        bti::bytes pfb(pf.begin(), pf.end());
                        bti::bytes rfb(rf.begin(), rf.end());
                        auto bp = bti::read_bti_partitions(pfb);
                        auto& pos = su.positions[s];
                        pos.reserve(bp.entries.size() + 1);
                        auto& pfx = su.bti_prefixes[s];
                        pfx.reserve(bp.entries.size());
                        for (auto& e : bp.entries) {
                            pos.push_back(e.idxpos < 0 ? (uint64_t)~e.idxpos
                                                       : bti::row_index_data_pos(rfb, (uint64_t)e.idxpos));
                            pfx.push_back(std::move(e.prefix));
                        }
                    } else {
                        su.index_data[s] = read_file(base + "-Index.db");
                    }
                    su.cinfos[s] = parse_compression_info(read_file(base + "-CompressionInfo.db"));
                    su.stats[s] = parse_statistics(read_file(base + "-Statistics.db"));
                    if (!preread_full) su.comp_file_sz[s] = file_size_of(base + "-Data.db");
                } catch (const std::exception& e) { merr[s] = e.what(); }
            });
        }
        for (auto& t : mth) t.join();
        for (int s = 0; s < k; s++)
            if (!merr[s].empty()) throw std::runtime_error(merr[s]);
    }
    for (int s = 0; s < k; s++) {
        const std::string& base = su.in_bases[s];
        auto& st = su.stats[s];
        if (st.regular_cols.empty() || st.regular_cols.size() > 63)
            throw std::runtime_error("1..63 regular columns supported");
        if (!st.partitioner.empty() && st.partitioner.find("Murmur3") == std::string::npos)
            throw std::runtime_error("Murmur3Partitioner required");
        size_t sl = base.rfind('/');
        std::string name = sl == std::string::npos ? base : base.substr(sl + 1);
        size_t a = name.find('-'), b2 = name.find('-', a + 1);
        su.generations[s] = std::stoull(name.substr(a + 1, b2 - a - 1));
    }
    {
        const auto& rcols = su.stats[0].regular_cols;
        for (size_t ci = 0; ci < rcols.size(); ci++) {
            const std::string& ct = rcols[ci].second;
            if (ct == "org.apache.cassandra.db.marshal.LongType") su.col_fixed_h.push_back(8);
            else if (ct == "org.apache.cassandra.db.marshal.Int32Type") su.col_fixed_h.push_back(4);
            else if (ct == "org.apache.cassandra.db.marshal.BytesType" ||
                     ct == "org.apache.cassandra.db.marshal.UTF8Type" ||
                     ct == "org.apache.cassandra.db.marshal.AsciiType")
                su.col_fixed_h.push_back(-1);
            else if (ct == "org.apache.cassandra.db.marshal.CounterColumnType") {
                su.col_fixed_h.push_back(-1);
                su.counters = true;
            }
            else if (ct == "org.apache.cassandra.db.marshal.MapType(org.apache.cassandra.db.marshal.BytesType,org.apache.cassandra.db.marshal.BytesType)") {
                // one complex column, and it must be the LAST regular column
                // (engine layout constraint; the generator names it 'zm' so
                // name order puts it last)
                if (su.n_cpx || ci + 1 != rcols.size())
                    throw std::runtime_error("at most one complex column, as the last regular column");
                su.n_cpx = 1;
            } else {
                throw std::runtime_error("unsupported column type " + ct);
            }
        }
        if (su.counters) {
            // counter tables hold only counter columns (CreateTableStatement
            // forbids mixing); complex/static columns are rejected with them
            if (su.col_fixed_h.size() != rcols.size() || su.n_cpx)
                throw std::runtime_error("counter tables must be all-counter columns");
            for (auto& cp : rcols)
                if (cp.second != "org.apache.cassandra.db.marshal.CounterColumnType")
                    throw std::runtime_error("counter tables must be all-counter columns");
            if (!su.stats[0].static_cols.empty())
                throw std::runtime_error("static columns with counters unsupported");
        }
    }
    for (auto& t : su.stats[0].clustering_types) su.ck_widths.push_back(ck_type_width(t));
    for (auto& [nm2, ct2] : su.stats[0].static_cols) {
        (void)nm2;
        if (ct2 == "org.apache.cassandra.db.marshal.LongType") su.static_fixed_h.push_back(8);
        else if (ct2 == "org.apache.cassandra.db.marshal.Int32Type") su.static_fixed_h.push_back(4);
        else if (ct2 == "org.apache.cassandra.db.marshal.BytesType" ||
                 ct2 == "org.apache.cassandra.db.marshal.UTF8Type" ||
                 ct2 == "org.apache.cassandra.db.marshal.AsciiType")
            su.static_fixed_h.push_back(-1);
        else throw std::runtime_error("unsupported static column type " + ct2);
    }
    if (su.static_fixed_h.size() > 63)
        throw std::runtime_error("1..63 static columns supported");
    if (su.ck_widths.size() > 32)
        throw std::runtime_error("at most 32 clustering columns supported");
    if (su.bti) {
        // positions collected from the tries above; append the end sentinel
        for (int s = 0; s < k; s++) su.positions[s].push_back(su.cinfos[s].data_len);
    } else {
        std::vector<std::thread> th;
        std::vector<std::string> perr(k);
        for (int s = 0; s < k; s++)
            th.emplace_back(parse_index_positions, std::cref(su.index_data[s]),
                            su.cinfos[s].data_len, std::ref(su.positions[s]),
                            preread_full ? nullptr : &su.entry_offs[s], std::ref(perr[s]));
        for (auto& t : th) t.join();
        for (int s = 0; s < k; s++)
            if (!perr[s].empty()) throw std::runtime_error("Index.db parse: " + perr[s]);
    }
    for (int s = 1; s < k; s++)
        if (su.cinfos[s].snappy != su.cinfos[0].snappy)
            throw std::runtime_error("mixed compressors across inputs unsupported");
}

// Murmur3 token of the key in Index.db entry at byte offset `off`
static int64_t index_entry_token(const bytes& ib, uint64_t off) {
    uint32_t klen = ((uint32_t)ib[off] << 8) | ib[off + 1];
    return murmur3_token(ib.data() + off + 2, klen);
}

extern "C" int gpuc_compact(const gpuc_job* job, gpuc_result* res) {
    memset(res, 0, sizeof(*res));
    double t_start_all;
    auto wall = []() {
        struct timespec ts;
        clock_gettime(CLOCK_MONOTONIC, &ts);
        return ts.tv_sec * 1e3 + ts.tv_nsec / 1e6;
    };
    t_start_all = wall();
    if (g_trace) g_trace_t0 = trace_wall();
    try {
        int ndev = gpuc_device_count();
        if (ndev <= 0) { set_err(res->error, sizeof(res->error), "no HIP device (no CPU fallback)"); return GPUC_ERR_NO_GPU; }
        HIP_CHECK(hipSetDevice(job->device));
        {
            hipStream_t s0;
            HIP_CHECK(hipStreamCreate(&s0));
            ensure_crc_tables(s0);
            HIP_CHECK(hipStreamDestroy(s0));
        }
        if (job->n_inputs < 1 || job->n_inputs + (job->n_tomb_sources > 0 ? job->n_tomb_sources : 0) > 64) {
            set_err(res->error, sizeof(res->error), "n_inputs (+ tombstone sources) must be 1..64");
            return GPUC_ERR_UNSUPPORTED;
        }
        double t0 = wall();
        int S_pre = job->n_output_shards > 1 ? job->n_output_shards : 1;
        CompactSetup su;
        compact_setup(job, su, S_pre == 1);
        res->ms_read_io = wall() - t0;
        TR("meta+index parsed");
        for (int s = 0; s < su.k_data; s++)
            res->input_uncompressed_bytes += su.cinfos[s].data_len;
        for (int s = 0; s < su.k_data; s++)
            res->partitions_in += su.positions[s].size() - 1;

        int S = S_pre;
        if (S > 1024) throw std::runtime_error("n_output_shards must be <= 1024");
        std::mutex res_mu;
        if (S == 1) {
            std::vector<std::pair<uint32_t, uint32_t>> pr(su.k);
            for (int s = 0; s < su.k; s++)
                pr[s] = {0u, (uint32_t)(su.positions[s].size() - 1)};
            compact_one(job, su, pr, job->output_base, 0, res, res_mu);
        } else {
            if (job->has_token_range)
                throw std::runtime_error("token_range + n_output_shards unsupported");
            // output bases: generation g, g+1, ... (a compaction may emit
            // several sstables — UCS shard model / SplittingCompactionWriter)
            std::string ob = job->output_base;
            size_t sl = ob.rfind('/');
            std::string dir = sl == std::string::npos ? std::string(".") : ob.substr(0, sl);
            std::string name = sl == std::string::npos ? ob : ob.substr(sl + 1);
            size_t a = name.find('-'), b2 = name.find('-', a + 1);
            uint64_t gen0 = std::stoull(name.substr(a + 1, b2 - a - 1));
            std::string pre = name.substr(0, a + 1), post = name.substr(b2);
            // equal token-range split (cassandra_amd/sharding.py contract)
            std::vector<int64_t> lo_tok(S + 1);
            for (int i = 0; i < S; i++) {
                uint64_t lo_u = (uint64_t)(((unsigned __int128)i << 64) / (unsigned)S);
                lo_tok[i] = (int64_t)(lo_u + 0x8000000000000000ULL);
            }
            // per input: shard boundary partition = first with token >= lo_tok[i]
            std::vector<std::vector<uint32_t>> bounds(su.k, std::vector<uint32_t>(S + 1));
            std::vector<std::vector<uint32_t>> boundsR(su.k, std::vector<uint32_t>(S + 1));
            for (int s = 0; s < su.k; s++) {
                uint32_t n = (uint32_t)(su.positions[s].size() - 1);
                bounds[s][0] = boundsR[s][0] = 0;
                bounds[s][S] = boundsR[s][S] = n;
                for (int i = 1; i < S; i++) {
                    if (!su.bti) {
                        uint32_t lo = 0, hi = n;  // lower_bound(token >= lo_tok[i])
                        while (lo < hi) {
                            uint32_t mid = (lo + hi) >> 1;
                            if (index_entry_token(su.index_data[s], su.entry_offs[s][mid]) < lo_tok[i])
                                lo = mid + 1;
                            else hi = mid;
                        }
                        bounds[s][i] = lo;
                    } else {
                        // da inputs: Partitions.db stores CUT byte-comparable
                        // separators S_j with key_{j-1} < S_j <= key_j, whose
                        // leading bytes are the sign-flipped BE token
                        // (bti_byte_comparable_m3). A cut separator can be a
                        // strict PREFIX of the boundary string B, which makes
                        // its side ambiguous — so each boundary gets a
                        // left-biased lower bound (prefix-of-B counts as > B;
                        // first hit is provably <= j0+1, where j0 = first
                        // partition with token >= the boundary, because
                        // S_{j0+1} > key_{j0} >= B is never a prefix of B)
                        // and a right-biased upper bound (prefix counts as
                        // <= B; every separator before j0 is < B, so the
                        // first hit is >= j0 and <= j0+1). Window i =
                        // [L[i]-1, R[i+1]) then OVERLAPS its neighbours by at
                        // most two partitions; compact_one's per-shard token
                        // filter (exact, on GPU-computed tokens) assigns each
                        // partition to exactly one shard.
                        // boundary string: NEXT_COMPONENT then the
                        // sign-flipped BE token (bti_byte_comparable_m3)
                        uint8_t B[9];
                        B[0] = 0x40;
                        uint64_t tb = (uint64_t)lo_tok[i] ^ (1ull << 63);
                        for (int b3 = 0; b3 < 8; b3++) B[1 + b3] = (uint8_t)(tb >> (8 * (7 - b3)));
                        auto search = [&](bool prefix_is_gt) {
                            uint32_t lo = 0, hi = n;
                            while (lo < hi) {
                                uint32_t mid = (lo + hi) >> 1;
                                const bti::bytes& p2 = su.bti_prefixes[s][mid];
                                size_t m2 = p2.size() < 9 ? p2.size() : 9;
                                int c2 = m2 ? memcmp(p2.data(), B, m2) : 0;
                                bool gt = c2 ? c2 > 0 : (p2.size() >= 9 ? false : prefix_is_gt);
                                if (gt) hi = mid;
                                else lo = mid + 1;
                            }
                            return lo;
                        };
                        uint32_t L = search(true);
                        boundsR[s][i] = search(false);
                        bounds[s][i] = L > 0 ? L - 1 : 0;
                    }
                }
            }
            // two workers: front-phase kernels of one shard overlap the
            // other shard's compress
            std::atomic<int> next{0};
            std::vector<std::string> werr(2);
            auto workfn = [&](int w) {
                for (;;) {
                    int i = next.fetch_add(1);
                    if (i >= S) break;
                    std::vector<std::pair<uint32_t, uint32_t>> pr(su.k);
                    for (int s = 0; s < su.k; s++) {
                        // bti: window = [left-biased start, right-biased end)
                        uint32_t e2 = su.bti ? boundsR[s][i + 1] : bounds[s][i + 1];
                        pr[s] = {std::min(bounds[s][i], e2), e2};
                    }
                    std::string out_i = dir + "/" + pre + std::to_string(gen0 + i) + post;
                    int64_t sh_lo = lo_tok[i];
                    int64_t sh_hi = i + 1 < S ? lo_tok[i + 1] - 1 : INT64_MAX;
                    try {
                        compact_one(job, su, pr, out_i, w, res, res_mu, su.bti, sh_lo, sh_hi);
                    } catch (const std::exception& e) {
                        werr[w] = e.what();
                        break;
                    }
                }
            };
            std::thread w0(workfn, 0), w1(workfn, 1);
            w0.join();
            w1.join();
            for (auto& e : werr)
                if (!e.empty()) {
                    if (e.find("cancel_flag set") != std::string::npos) throw cancelled_error();
                    throw std::runtime_error(e);
                }
        }
        res->ms_total = wall() - t_start_all;
        return GPUC_OK;
    } catch (const cancelled_error& e) {
        set_err(res->error, sizeof(res->error), e.what());
        res->ms_total = wall() - t_start_all;
        return GPUC_ERR_CANCELLED;
    } catch (const std::exception& e) {
        set_err(res->error, sizeof(res->error), e.what());
        res->ms_total = wall() - t_start_all;
        return GPUC_ERR_INTERNAL;
    }
}


// One-sstable verification (Verifier.java / `nodetool verify --extended`
// semantics): CompressionInfo + per-chunk CRC32 (verified inside the
// decompress kernel), full row-format walk (parse pass A validates structure
// and sizes), strict DecoratedKey order, Digest.crc32 recomputed from the
// chunk frames, and the bloom filter rebuilt from the keys and compared with
// Filter.db. Runs the same GPU ingest path as compaction.
extern "C" int gpuc_verify(const char* input_base, int32_t device, char* error,
                           size_t error_len) {
    try {
        if (gpuc_device_count() <= 0) { set_err(error, error_len, "no HIP device"); return GPUC_ERR_NO_GPU; }
        HIP_CHECK(hipSetDevice(device));
        hipStream_t stream;
        HIP_CHECK(hipStreamCreate(&stream));
        ensure_crc_tables(stream);
        std::string base = input_base;
        bytes index_data = read_file(base + "-Index.db");
        HCompressionInfo ci = parse_compression_info(read_file(base + "-CompressionInfo.db"));
        HStatistics st = parse_statistics(read_file(base + "-Statistics.db"));
        size_t comp_sz = file_size_of(base + "-Data.db");
        bytes comp = read_file(base + "-Data.db");
        std::vector<uint64_t> positions;
        std::string perr;
        parse_index_positions(index_data, ci.data_len, positions, nullptr, perr);
        if (!perr.empty()) throw std::runtime_error("Index.db: " + perr);
        uint64_t n_parts = positions.size() - 1;

        // digest from the chunk frames (CRC fields validated against bytes below)
        {
            static Crc32Combiner comb;
            uint32_t tab[256];
            crc32_make_table(tab);
            uint32_t digest = 0;
            for (size_t c = 0; c < ci.offsets.size(); c++) {
                uint64_t off = ci.offsets[c];
                uint64_t end = c + 1 < ci.offsets.size() ? ci.offsets[c + 1] : comp_sz;
                uint64_t blen = end - off - 4;
                uint32_t ccrc = ((uint32_t)comp[end - 4] << 24) | ((uint32_t)comp[end - 3] << 16) |
                                ((uint32_t)comp[end - 2] << 8) | comp[end - 1];
                digest = comb.combine(digest, ccrc, blen);
                digest = comb.combine(digest, crc32_update_t(0, comp.data() + end - 4, 4, tab), 4);
            }
            bytes dg = read_file(base + "-Digest.crc32");
            std::string want(dg.begin(), dg.end());
            if (want != std::to_string(digest))
                throw std::runtime_error("Digest.crc32 mismatch");
        }

        // GPU: decompress (chunk CRC verify) + structural walk + order + bloom
        DevBuf d_comp, d_data, d_pos, d_chunks, d_error, d_recs;
        d_comp.alloc(comp.size());
        HIP_CHECK(hipMemcpyAsync(d_comp.p, comp.data(), comp.size(), hipMemcpyHostToDevice, stream));
        d_data.alloc(ci.data_len + 16);
        d_pos.alloc(positions.size() * 8);
        HIP_CHECK(hipMemcpyAsync(d_pos.p, positions.data(), positions.size() * 8,
                                 hipMemcpyHostToDevice, stream));
        std::vector<ChunkDesc> chunks;
        for (size_t c = 0; c < ci.offsets.size(); c++) {
            uint64_t off = ci.offsets[c];
            uint64_t end = c + 1 < ci.offsets.size() ? ci.offsets[c + 1] : comp_sz;
            ChunkDesc cd;
            cd.comp = d_comp.as<uint8_t>() + off;
            cd.out = d_data.as<uint8_t>() + c * (uint64_t)CHUNK_LEN;
            cd.comp_len = (uint32_t)(end - off - 4);
            cd.out_len = (uint32_t)std::min<uint64_t>(CHUNK_LEN, ci.data_len - c * (uint64_t)CHUNK_LEN);
            chunks.push_back(cd);
        }
        d_chunks.alloc(chunks.size() * sizeof(ChunkDesc) + 16);
        HIP_CHECK(hipMemcpyAsync(d_chunks.p, chunks.data(), chunks.size() * sizeof(ChunkDesc),
                                 hipMemcpyHostToDevice, stream));
        d_error.alloc(8);
        HIP_CHECK(hipMemsetAsync(d_error.p, 0, 8, stream));
        if (!chunks.empty())
            if (ci.snappy)
                hipLaunchKernelGGL(k_snappy_decompress_chunks,
                                   dim3(lz4_decomp_grid((uint32_t)chunks.size(), 1)), dim3(WAVE), 0,
                                   stream, d_chunks.as<ChunkDesc>(), (uint32_t)chunks.size(), 1,
                                   d_error.as<unsigned long long>(), (const uint32_t*)g_crc256,
                                   (uint8_t*)nullptr);
            else
                hipLaunchKernelGGL(k_lz4_decompress_wave, dim3(lz4_decomp_grid((uint32_t)chunks.size(), 1)), dim3(WAVE), 0,
                               stream, d_chunks.as<ChunkDesc>(), (uint32_t)chunks.size(), 1,
                               d_error.as<unsigned long long>(), (const uint32_t*)g_crc256);
        {
            unsigned long long e = 0;
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(&e, d_error.p, 8, hipMemcpyDeviceToHost));
            if (e == 1) throw std::runtime_error("chunk CRC mismatch");
            if (e) throw std::runtime_error("chunk decompress failed, code " + std::to_string(e));
        }
        SchemaParams sch{};
        std::vector<int32_t> cfh;
        for (auto& [nm, ct] : st.regular_cols) {
            (void)nm;
            if (ct == "org.apache.cassandra.db.marshal.LongType") cfh.push_back(8);
            else if (ct == "org.apache.cassandra.db.marshal.Int32Type") cfh.push_back(4);
            else if (ct == "org.apache.cassandra.db.marshal.MapType(org.apache.cassandra.db.marshal.BytesType,org.apache.cassandra.db.marshal.BytesType)") sch.n_cpx = 1;  // last regular column
            else cfh.push_back(-1);
        }
        if (cfh.empty() && !sch.n_cpx) cfh.push_back(-1);
        sch.n_cols = (uint32_t)cfh.size();
        if (!sch.n_cols && !sch.n_cpx) sch.n_cols = 1;
        DevBuf d_cf;
        d_cf.alloc(cfh.size() * 4);
        HIP_CHECK(hipMemcpyAsync(d_cf.p, cfh.data(), cfh.size() * 4, hipMemcpyHostToDevice, stream));
        sch.col_fixed = d_cf.as<int32_t>();
        std::vector<int32_t> vckw;
        for (auto& ct : st.clustering_types) vckw.push_back(ck_type_width(ct));
        sch.n_ck = (uint32_t)vckw.size();
        DevBuf d_vckw;
        d_vckw.alloc(vckw.size() * 4 + 8);
        if (sch.n_ck)
            HIP_CHECK(hipMemcpyAsync(d_vckw.p, vckw.data(), vckw.size() * 4,
                                     hipMemcpyHostToDevice, stream));
        sch.ck_w = d_vckw.as<int32_t>();
        std::vector<int32_t> vsf;
        for (auto& [snm, sct] : st.static_cols) {
            (void)snm;
            if (sct == "org.apache.cassandra.db.marshal.LongType") vsf.push_back(8);
            else if (sct == "org.apache.cassandra.db.marshal.Int32Type") vsf.push_back(4);
            else vsf.push_back(-1);
        }
        sch.n_static = (uint32_t)vsf.size();
        DevBuf d_vsf;
        d_vsf.alloc(vsf.size() * 4 + 8);
        if (sch.n_static)
            HIP_CHECK(hipMemcpyAsync(d_vsf.p, vsf.data(), vsf.size() * 4,
                                     hipMemcpyHostToDevice, stream));
        sch.static_fixed = d_vsf.as<int32_t>();
        sch.column_index_size = 64 * 1024;
        SrcDesc2 src{};
        src.data = d_data.as<uint8_t>();
        src.part_pos = d_pos.as<uint64_t>();
        src.n_parts = (uint32_t)n_parts;
        src.min_ts = st.hdr_min_ts;
        src.min_ldt = st.hdr_min_ldt;
        src.min_ttl = st.hdr_min_ttl;
        src.rec_base = 0;
        DevBuf d_src;
        d_src.alloc(sizeof(src));
        HIP_CHECK(hipMemcpyAsync(d_src.p, &src, sizeof(src), hipMemcpyHostToDevice, stream));
        d_recs.alloc((n_parts + 1) * sizeof(MRec));
        DevBuf p_pdm, p_pdl, p_rcnt, p_rbase, p_kaddr;
        ParsedCols pc{};
        p_pdm.alloc(n_parts * 8 + 8); pc.pdel_mfda = p_pdm.as<int64_t>();
        p_pdl.alloc(n_parts * 4 + 8); pc.pdel_ldt = p_pdl.as<uint32_t>();
        p_rcnt.alloc(n_parts * 4 + 8); pc.row_count = p_rcnt.as<uint32_t>();
        p_rbase.alloc(n_parts * 8 + 8); pc.row_base = p_rbase.as<uint64_t>();
        p_kaddr.alloc(n_parts * 8 + 8); pc.key_addr = p_kaddr.as<uint64_t>();
        uint32_t blocks = (uint32_t)((n_parts + 255) / 256);
        if (n_parts) {
            hipLaunchKernelGGL(k_parse_count, dim3(blocks), dim3(256), 0, stream,
                               d_src.as<SrcDesc2>(), 1u, (uint32_t)n_parts, d_recs.as<MRec>(),
                               pc, sch, d_error.as<unsigned long long>());
            KeyLut lut{};
            lut.base[0] = d_data.as<uint8_t>();
            lut.pos[0] = d_pos.as<uint64_t>();
            lut.enabled = 1;
            hipLaunchKernelGGL(k_verify_order, dim3(blocks), dim3(256), 0, stream,
                               d_recs.as<MRec>(), n_parts, lut, d_error.as<unsigned long long>());
        }
        // bloom recheck
        {
            bytes f = read_file(base + "-Filter.db");
            HReader r(f);
            uint32_t bk = r.be32();
            uint32_t words = r.be32();
            DevBuf d_bloom;
            d_bloom.alloc((uint64_t)words * 8 + 8);
            HIP_CHECK(hipMemsetAsync(d_bloom.p, 0, (uint64_t)words * 8, stream));
            if (n_parts)
                hipLaunchKernelGGL(k_verify_bloom, dim3(blocks), dim3(256), 0, stream, pc,
                                   d_recs.as<MRec>(), n_parts, d_bloom.as<uint32_t>(),
                                   (uint64_t)words * 64, (int32_t)bk);
            std::vector<uint8_t> got(words * 8);
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(got.data(), d_bloom.p, words * 8, hipMemcpyDeviceToHost));
            unsigned long long gerr = 0;
            HIP_CHECK(hipMemcpy(&gerr, d_error.p, 8, hipMemcpyDeviceToHost));
            if (gerr == 1) throw std::runtime_error("chunk CRC mismatch");
            if (gerr == 30) throw std::runtime_error("partitions out of order");
            if (gerr) throw std::runtime_error("row format walk failed, code " + std::to_string(gerr));
            if (memcmp(got.data(), f.data() + 8, std::min<size_t>(got.size(), f.size() - 8)) != 0)
                throw std::runtime_error("Filter.db does not match keys");
        }
        HIP_CHECK(hipStreamDestroy(stream));
        return GPUC_OK;
    } catch (const std::exception& e) {
        set_err(error, error_len, e.what());
        return GPUC_ERR_FORMAT;
    }
}

// Scrub (SortedTableScrubber semantics, this implementation's recovery
// granularity — mirrored by oracle::scrub_sstable): salvage partitions whose
// byte range touches only CRC/decode-clean chunks, rewrite them through the
// standard writer; report kept/dropped counts.
extern "C" int gpuc_scrub(const char* input_base, const char* output_base, int32_t device,
                          uint64_t* kept, uint64_t* dropped, char* error, size_t error_len) {
    try {
        if (gpuc_device_count() <= 0) { set_err(error, error_len, "no HIP device"); return GPUC_ERR_NO_GPU; }
        HIP_CHECK(hipSetDevice(device));
        hipStream_t stream;
        HIP_CHECK(hipStreamCreate(&stream));
        ensure_crc_tables(stream);
        std::string base = input_base;
        bytes index_data = read_file(base + "-Index.db");
        HCompressionInfo ci = parse_compression_info(read_file(base + "-CompressionInfo.db"));
        HStatistics st = parse_statistics(read_file(base + "-Statistics.db"));
        size_t comp_sz = file_size_of(base + "-Data.db");
        bytes comp = read_file(base + "-Data.db");
        std::vector<uint64_t> positions;
        std::string perr;
        parse_index_positions(index_data, ci.data_len, positions, nullptr, perr);
        if (!perr.empty()) throw std::runtime_error("Index.db: " + perr);
        uint64_t n_parts = positions.size() - 1;
        size_t n_chunks = ci.offsets.size();

        // decompress with per-chunk bad flags
        DevBuf d_comp, d_data, d_chunksb, d_error, d_bad;
        d_comp.alloc(comp.size());
        HIP_CHECK(hipMemcpyAsync(d_comp.p, comp.data(), comp.size(), hipMemcpyHostToDevice, stream));
        d_data.alloc(ci.data_len + 16);
        d_bad.alloc(n_chunks + 8);
        HIP_CHECK(hipMemsetAsync(d_bad.p, 0, n_chunks + 8, stream));
        d_error.alloc(8);
        HIP_CHECK(hipMemsetAsync(d_error.p, 0, 8, stream));
        std::vector<ChunkDesc> chunks;
        for (size_t c = 0; c < n_chunks; c++) {
            uint64_t off = ci.offsets[c];
            uint64_t end = c + 1 < n_chunks ? ci.offsets[c + 1] : comp_sz;
            ChunkDesc cd;
            cd.comp = d_comp.as<uint8_t>() + off;
            cd.out = d_data.as<uint8_t>() + c * (uint64_t)CHUNK_LEN;
            cd.comp_len = (uint32_t)(end - off - 4);
            cd.out_len = (uint32_t)std::min<uint64_t>(CHUNK_LEN, ci.data_len - c * (uint64_t)CHUNK_LEN);
            chunks.push_back(cd);
        }
        d_chunksb.alloc(chunks.size() * sizeof(ChunkDesc) + 16);
        HIP_CHECK(hipMemcpyAsync(d_chunksb.p, chunks.data(), chunks.size() * sizeof(ChunkDesc),
                                 hipMemcpyHostToDevice, stream));
        if (!chunks.empty())
            if (ci.snappy)
                hipLaunchKernelGGL(k_snappy_decompress_chunks,
                                   dim3(lz4_decomp_grid((uint32_t)chunks.size(), 1)), dim3(WAVE), 0,
                                   stream, d_chunksb.as<ChunkDesc>(), (uint32_t)chunks.size(), 1,
                                   d_error.as<unsigned long long>(), (const uint32_t*)g_crc256,
                                   d_bad.as<uint8_t>());
            else
                hipLaunchKernelGGL(k_lz4_decompress_wave, dim3(lz4_decomp_grid((uint32_t)chunks.size(), 1)), dim3(WAVE), 0,
                               stream, d_chunksb.as<ChunkDesc>(), (uint32_t)chunks.size(), 1,
                               d_error.as<unsigned long long>(), (const uint32_t*)g_crc256,
                               d_bad.as<uint8_t>());
        std::vector<uint8_t> bad(n_chunks);
        HIP_CHECK(hipStreamSynchronize(stream));
        if (n_chunks)
            HIP_CHECK(hipMemcpy(bad.data(), d_bad.p, n_chunks, hipMemcpyDeviceToHost));

        // kept partitions: all overlapped chunks clean
        std::vector<uint64_t> kpos, kend;
        for (uint64_t i = 0; i < n_parts; i++) {
            uint64_t c0 = positions[i] / CHUNK_LEN;
            uint64_t c1 = (positions[i + 1] + CHUNK_LEN - 1) / CHUNK_LEN;
            bool ok = true;
            for (uint64_t c = c0; c < c1 && c < n_chunks; c++)
                if (bad[c]) ok = false;
            if (ok) {
                kpos.push_back(positions[i]);
                kend.push_back(positions[i + 1]);
            }
        }
        uint64_t n_kept = kpos.size();
        if (kept) *kept = n_kept;
        if (dropped) *dropped = n_parts - n_kept;

        // schema (same resolution as compaction)
        SchemaParams sch{};
        std::vector<int32_t> cfh, vckw, vsf;
        for (auto& [nm, ct] : st.regular_cols) {
            (void)nm;
            if (ct == "org.apache.cassandra.db.marshal.MapType(org.apache.cassandra.db.marshal.BytesType,org.apache.cassandra.db.marshal.BytesType)") sch.n_cpx = 1;  // last regular column
            else if (ct == "org.apache.cassandra.db.marshal.CounterColumnType") {
                cfh.push_back(-1);
                sch.counters = 1;
            } else cfh.push_back(ck_type_width(ct));
        }
        for (auto& ct : st.clustering_types) vckw.push_back(ck_type_width(ct));
        for (auto& [nm, ct] : st.static_cols) { (void)nm; vsf.push_back(ck_type_width(ct)); }
        sch.n_cols = (uint32_t)cfh.size();
        sch.n_ck = (uint32_t)vckw.size();
        sch.n_static = (uint32_t)vsf.size();
        DevBuf d_cf, d_ckw, d_sf;
        d_cf.alloc(cfh.size() * 4 + 8);
        if (!cfh.empty()) HIP_CHECK(hipMemcpyAsync(d_cf.p, cfh.data(), cfh.size() * 4, hipMemcpyHostToDevice, stream));
        d_ckw.alloc(vckw.size() * 4 + 8);
        if (!vckw.empty()) HIP_CHECK(hipMemcpyAsync(d_ckw.p, vckw.data(), vckw.size() * 4, hipMemcpyHostToDevice, stream));
        d_sf.alloc(vsf.size() * 4 + 8);
        if (!vsf.empty()) HIP_CHECK(hipMemcpyAsync(d_sf.p, vsf.data(), vsf.size() * 4, hipMemcpyHostToDevice, stream));
        sch.col_fixed = d_cf.as<int32_t>();
        sch.ck_w = d_ckw.as<int32_t>();
        sch.static_fixed = d_sf.as<int32_t>();
        sch.column_index_size = 64 * 1024;

        // parse kept partitions (explicit ends: gaps where partitions dropped)
        DevBuf d_kpos, d_kend, d_recs, d_rows_in;
        d_kpos.alloc(n_kept * 8 + 8);
        d_kend.alloc(n_kept * 8 + 8);
        if (n_kept) {
            HIP_CHECK(hipMemcpyAsync(d_kpos.p, kpos.data(), n_kept * 8, hipMemcpyHostToDevice, stream));
            HIP_CHECK(hipMemcpyAsync(d_kend.p, kend.data(), n_kept * 8, hipMemcpyHostToDevice, stream));
        }
        SrcDesc2 src{};
        src.data = d_data.as<uint8_t>();
        src.part_pos = d_kpos.as<uint64_t>();
        src.part_end = d_kend.as<uint64_t>();
        src.n_parts = (uint32_t)n_kept;
        src.min_ts = st.hdr_min_ts;
        src.min_ldt = st.hdr_min_ldt;
        src.min_ttl = st.hdr_min_ttl;
        src.rec_base = 0;
        DevBuf d_src;
        d_src.alloc(sizeof(src));
        HIP_CHECK(hipMemcpyAsync(d_src.p, &src, sizeof(src), hipMemcpyHostToDevice, stream));
        d_recs.alloc(n_kept * sizeof(MRec) + 32);
        d_rows_in.alloc(8);
        HIP_CHECK(hipMemsetAsync(d_rows_in.p, 0, 8, stream));
        DevBuf p_pdm, p_pdl, p_rcnt, p_rbase, p_kaddr;
        ParsedCols pc{};
        p_pdm.alloc(n_kept * 8 + 8); pc.pdel_mfda = p_pdm.as<int64_t>();
        p_pdl.alloc(n_kept * 4 + 8); pc.pdel_ldt = p_pdl.as<uint32_t>();
        p_rcnt.alloc(n_kept * 4 + 8); pc.row_count = p_rcnt.as<uint32_t>();
        p_rbase.alloc(n_kept * 8 + 8); pc.row_base = p_rbase.as<uint64_t>();
        p_kaddr.alloc(n_kept * 8 + 8); pc.key_addr = p_kaddr.as<uint64_t>();
        StaticColsBuf p_static;
        p_static.alloc(sch.n_static ? n_kept : 1, sch.n_static);
        pc.st = p_static.st;
        uint32_t blocks = (uint32_t)((n_kept + 255) / 256);
        if (n_kept) {
            hipLaunchKernelGGL(k_parse_count, dim3(blocks), dim3(256), 0, stream,
                               d_src.as<SrcDesc2>(), 1u, (uint32_t)n_kept, d_recs.as<MRec>(), pc,
                               sch, d_error.as<unsigned long long>());
            hipLaunchKernelGGL(k_widen_u32, dim3(blocks), dim3(256), 0, stream, pc.row_count,
                               pc.row_base, n_kept);
        }
        uint64_t total_rows = n_kept ? exscan_u64(pc.row_base, n_kept, stream) : 0;
        UnfColsBuf in_rows;
        in_rows.alloc(total_rows, sch.n_cols, sch.n_ck, sch.n_cpx);
        DevBuf p_cpxtot, p_cpxbase;
        uint64_t total_in_cpx = 0;
        if (n_kept && sch.n_cpx) {
            // complex-cell counting pass, then arena alloc (compact's B1/B2)
            p_cpxtot.alloc(n_kept * 4 + 8);
            pc.cpx_total = p_cpxtot.as<uint32_t>();
            hipLaunchKernelGGL(k_parse_rows, dim3(blocks), dim3(256), 0, stream,
                               d_src.as<SrcDesc2>(), 1u, (uint32_t)n_kept, pc, in_rows.uc, sch,
                               d_error.as<unsigned long long>(),
                               d_rows_in.as<unsigned long long>(), nullptr);
            HIP_CHECK(hipMemsetAsync(d_rows_in.p, 0, 8, stream));
            p_cpxbase.alloc(n_kept * 8 + 8);
            hipLaunchKernelGGL(k_widen_u32, dim3(blocks), dim3(256), 0, stream, pc.cpx_total,
                               p_cpxbase.as<uint64_t>(), n_kept);
            total_in_cpx = exscan_u64(p_cpxbase.as<uint64_t>(), n_kept, stream);
            in_rows.alloc_cpx_arena(total_in_cpx);
        }
        if (n_kept)
            hipLaunchKernelGGL(k_parse_rows, dim3(blocks), dim3(256), 0, stream,
                               d_src.as<SrcDesc2>(), 1u, (uint32_t)n_kept, pc, in_rows.uc, sch,
                               d_error.as<unsigned long long>(), d_rows_in.as<unsigned long long>(),
                               sch.n_cpx ? p_cpxbase.as<uint64_t>() : nullptr);
        {
            unsigned long long e = 0;
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(&e, d_error.p, 8, hipMemcpyDeviceToHost));
            if (e) throw std::runtime_error("scrub parse failed, code " + std::to_string(e));
        }
        // single source: records are already in DecoratedKey order
        KeyLut lut{};
        lut.base[0] = d_data.as<uint8_t>();
        lut.pos[0] = d_kpos.as<uint64_t>();
        lut.enabled = 1;
        DevBuf d_head, d_gstart, d_ng;
        d_head.alloc(n_kept * 8 + 8);
        d_gstart.alloc(n_kept * 8 + 8);
        d_ng.alloc(8);
        HIP_CHECK(hipMemsetAsync(d_ng.p, 0, 8, stream));
        if (n_kept) {
            hipLaunchKernelGGL(k_group_heads, dim3(blocks), dim3(256), 0, stream, d_recs.as<MRec>(),
                               n_kept, d_head.as<uint64_t>(), lut);
            exscan_u64(d_head.as<uint64_t>(), n_kept, stream);
            hipLaunchKernelGGL(k_group_starts2, dim3(blocks), dim3(256), 0, stream,
                               d_recs.as<MRec>(), n_kept, d_head.as<uint64_t>(),
                               d_gstart.as<uint64_t>(), d_ng.as<uint64_t>(), lut);
        }
        uint64_t n_groups = 0;
        HIP_CHECK(hipStreamSynchronize(stream));
        HIP_CHECK(hipMemcpy(&n_groups, d_ng.p, 8, hipMemcpyDeviceToHost));
        DevBuf d_srcbases, d_grows, d_stats, d_tomb;
        uint32_t zero = 0;
        d_srcbases.alloc(8);
        HIP_CHECK(hipMemcpyAsync(d_srcbases.p, &zero, 4, hipMemcpyHostToDevice, stream));
        d_grows.alloc(n_groups * 8 + 8);
        if (n_groups) {
            uint32_t gblocks = (uint32_t)((n_groups + 255) / 256);
            hipLaunchKernelGGL(k_group_row_sums, dim3(gblocks), dim3(256), 0, stream,
                               d_recs.as<MRec>(), d_gstart.as<uint64_t>(), n_groups, n_kept,
                               d_srcbases.as<uint32_t>(), pc, d_grows.as<uint64_t>());
        }
        uint64_t out_total = n_groups ? exscan_u64(d_grows.as<uint64_t>(), n_groups, stream) : 0;
        OutPartsBuf opb;
        opb.alloc(n_groups ? n_groups : 1, sch.n_static);
        UnfColsBuf out_rows;
        out_rows.alloc(out_total, sch.n_cols, sch.n_ck, sch.n_cpx);
        if (sch.n_cpx) out_rows.alloc_cpx_arena(total_in_cpx);
        DevBuf d_ctr_arena_s;
        if (sch.counters) {
            // merged-context arena (arity-1 reconcile still routes counter
            // cells through the context branch)
            DevBuf d_sum;
            d_sum.alloc(8);
            HIP_CHECK(hipMemsetAsync(d_sum.p, 0, 8, stream));
            uint64_t n_cells = total_rows * sch.n_cols;
            if (n_cells) {
                uint32_t blocks2 = (uint32_t)((n_cells + 255) / 256);
                hipLaunchKernelGGL(k_sum_vallen, dim3(blocks2), dim3(256), 0, stream,
                                   in_rows.uc, n_cells, d_sum.as<unsigned long long>());
            }
            unsigned long long total_ctr = 0;
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(&total_ctr, d_sum.p, 8, hipMemcpyDeviceToHost));
            d_ctr_arena_s.alloc(total_ctr + 64);
            sch.ctr_arena = d_ctr_arena_s.as<uint8_t>();
        }
        d_stats.alloc(sizeof(OutStats));
        init_outstats(d_stats, stream);
        uint32_t tomb_cap = (uint32_t)std::min<uint64_t>(
            out_total * ((uint64_t)sch.n_cols + sch.n_cpx + 2) + total_in_cpx +
                n_groups * ((uint64_t)sch.n_static + 2) + 1024,
            400000000ull);
        d_tomb.alloc((uint64_t)tomb_cap * 4);
        PurgeParams2 pp{};
        pp.gc_before = INT64_MIN;
        pp.never_purge = 1;
        if (n_groups) {
            uint32_t gblocks = (uint32_t)((n_groups + 255) / 256);
            hipLaunchKernelGGL(k_reconcile_rows<8>, dim3(gblocks), dim3(256), 0, stream,
                               d_recs.as<MRec>(), d_gstart.as<uint64_t>(), n_groups, n_kept,
                               d_srcbases.as<uint32_t>(), pc, in_rows.uc, opb.op, out_rows.uc,
                               d_grows.as<uint64_t>(), sch, pp, d_stats.as<OutStats>(),
                               d_error.as<unsigned long long>());
        }
        HIP_CHECK(hipStreamSynchronize(stream));
        {
            unsigned long long e = 0;
            HIP_CHECK(hipMemcpy(&e, d_error.p, 8, hipMemcpyDeviceToHost));
            if (e) throw std::runtime_error("scrub reconcile failed, code " + std::to_string(e));
        }
        SerParams2 sp2{};
        sp2.sch = sch;
        sp2.hs.min_ts = st.min_timestamp == NO_TIMESTAMP ? TIMESTAMP_EPOCH : st.min_timestamp;
        sp2.hs.min_ldt = st.min_ldt == NO_DELETION_TIME ? DELETION_TIME_EPOCH : st.min_ldt;
        sp2.hs.min_ttl = st.min_ttl == INT32_MAX ? 0 : st.min_ttl;
        write_sstable_device(opb, out_rows, n_groups, sp2, d_stats, d_tomb, tomb_cap, output_base,
                             st.key_type, st.clustering_types, st.regular_cols, st.static_cols,
                             stream, 0, ci.snappy);
        HIP_CHECK(hipStreamDestroy(stream));
        return GPUC_OK;
    } catch (const std::exception& e) {
        set_err(error, error_len, e.what());
        return GPUC_ERR_INTERNAL;
    }
}

// Memtable flush: unsorted unique-key rows from the host ->
// token sort on device -> the shared serialize/compress/index/bloom kernels
// -> one complete `oa` sstable. Schema: `pk blob PRIMARY KEY, val blob`
// (one regular column; row tombstones via del_ldt != UINT32_MAX).
extern "C" int gpuc_flush(const gpuc_flush_rows* rows, const char* output_base,
                          int32_t device, char* error, size_t error_len) {
    try {
        if (gpuc_device_count() <= 0) { set_err(error, error_len, "no HIP device"); return GPUC_ERR_NO_GPU; }
        if (!rows || rows->n_rows == 0) { set_err(error, error_len, "no rows"); return GPUC_ERR_UNSUPPORTED; }
        HIP_CHECK(hipSetDevice(device));
        hipStream_t stream;
        HIP_CHECK(hipStreamCreate(&stream));
        ensure_crc_tables(stream);
        uint64_t n = rows->n_rows;
        // flatten to arenas
        std::vector<uint64_t> koff(n + 1, 0), voff(n + 1, 0);
        for (uint64_t i = 0; i < n; i++) {
            if (rows->key_lens[i] == 0) throw std::runtime_error("empty key");
            koff[i + 1] = koff[i] + rows->key_lens[i];
            voff[i + 1] = voff[i] + (rows->values[i] ? rows->value_lens[i] : 0);
        }
        std::vector<uint8_t> kbuf(koff[n]), vbuf(voff[n] ? voff[n] : 1);
        std::vector<uint32_t> dldt(n);
        for (uint64_t i = 0; i < n; i++) {
            memcpy(kbuf.data() + koff[i], rows->keys[i], rows->key_lens[i]);
            if (rows->values[i]) {
                memcpy(vbuf.data() + voff[i], rows->values[i], rows->value_lens[i]);
                dldt[i] = LDT_NONE_U32;
            } else {
                dldt[i] = rows->del_ldts[i];
                if (dldt[i] == LDT_NONE_U32)
                    throw std::runtime_error("row without value needs del_ldt");
            }
        }
        DevBuf d_k, d_koff, d_v, d_voff, d_ts, d_dldt, d_a, d_b, d_err;
        d_k.alloc(kbuf.size()); d_koff.alloc((n + 1) * 8);
        d_v.alloc(vbuf.size()); d_voff.alloc((n + 1) * 8);
        d_ts.alloc(n * 8); d_dldt.alloc(n * 4);
        HIP_CHECK(hipMemcpyAsync(d_k.p, kbuf.data(), kbuf.size(), hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(d_koff.p, koff.data(), (n + 1) * 8, hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(d_v.p, vbuf.data(), vbuf.size(), hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(d_voff.p, voff.data(), (n + 1) * 8, hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(d_ts.p, rows->timestamps, n * 8, hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(d_dldt.p, dldt.data(), n * 4, hipMemcpyHostToDevice, stream));
        FlushParams fp{};
        fp.keys = d_k.as<uint8_t>();
        fp.key_off = d_koff.as<uint64_t>();
        fp.ts = d_ts.as<int64_t>();
        fp.vals = d_v.as<uint8_t>();
        fp.val_off = d_voff.as<uint64_t>();
        fp.del_ldt = d_dldt.as<uint32_t>();
        fp.n = n;
        d_a.alloc(n * sizeof(MRec));
        d_b.alloc(n * sizeof(MRec));
        uint32_t blocks = (uint32_t)((n + 255) / 256);
        hipLaunchKernelGGL(k_flush_recs, dim3(blocks), dim3(256), 0, stream, fp, d_a.as<MRec>());
        // exact-key comparator over the key arena (offsets pre-shifted by -2
        // to cancel the short-length skip in mrec_key_bytes)
        std::vector<uint64_t> koff_adj(n);
        for (uint64_t i = 0; i < n; i++) koff_adj[i] = koff[i] - 2;
        DevBuf d_koff_adj;
        d_koff_adj.alloc(n * 8);
        HIP_CHECK(hipMemcpyAsync(d_koff_adj.p, koff_adj.data(), n * 8, hipMemcpyHostToDevice, stream));
        KeyLut lut{};
        lut.base[0] = d_k.as<uint8_t>();
        lut.pos[0] = d_koff_adj.as<uint64_t>();
        lut.enabled = 1;
        MRec* d_sorted = merge_sort_recs(d_a.as<MRec>(), d_b.as<MRec>(), n, stream, lut);
        d_err.alloc(8);
        HIP_CHECK(hipMemsetAsync(d_err.p, 0, 8, stream));
        hipLaunchKernelGGL(k_flush_dupcheck, dim3(blocks), dim3(256), 0, stream, fp, d_sorted,
                           d_err.as<unsigned long long>());
        OutPartsBuf opb;
        opb.alloc(n);
        UnfColsBuf urows;
        urows.alloc(n, 1, 0);
        hipLaunchKernelGGL(k_flush_fill, dim3(blocks), dim3(256), 0, stream, fp, d_sorted,
                           opb.op, urows.uc);
        {
            unsigned long long e = 0;
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(&e, d_err.p, 8, hipMemcpyDeviceToHost));
            if (e == 40) throw std::runtime_error("duplicate partition keys in flush input");
        }
        // header mins (SerializationHeader.make at flush: collect, then write)
        DevBuf d_stats, d_tomb;
        d_stats.alloc(sizeof(OutStats));
        init_outstats(d_stats, stream);
        uint32_t tomb_cap = (uint32_t)std::min<uint64_t>(n * 3 + 1024, 400000000ull);
        d_tomb.alloc((uint64_t)tomb_cap * 4);
        hipLaunchKernelGGL(k_collect_rows, dim3(blocks), dim3(256), 0, stream, opb.op,
                           urows.uc, n, 1u, d_stats.as<OutStats>(), d_tomb.as<uint32_t>(), tomb_cap);
        OutStats hs0;
        HIP_CHECK(hipStreamSynchronize(stream));
        HIP_CHECK(hipMemcpy(&hs0, d_stats.p, sizeof(OutStats), hipMemcpyDeviceToHost));
        SerParams2 sp{};
        sp.hs.min_ts = hs0.min_ts_flip == 0xFFFFFFFFFFFFFFFFULL
                           ? TIMESTAMP_EPOCH : (int64_t)(hs0.min_ts_flip ^ 0x8000000000000000ULL);
        sp.hs.min_ldt = hs0.min_ldt_flip == 0xFFFFFFFFFFFFFFFFULL
                            ? DELETION_TIME_EPOCH : (int64_t)(hs0.min_ldt_flip ^ 0x8000000000000000ULL);
        if (sp.hs.min_ldt == NO_DELETION_TIME) sp.hs.min_ldt = DELETION_TIME_EPOCH;
        sp.hs.min_ttl = 0;
        sp.sch.n_ck = 0;
        DevBuf d_ckw;
        d_ckw.alloc(8);
        sp.sch.ck_w = d_ckw.as<int32_t>();
        sp.sch.n_cols = 1;
        std::vector<int32_t> cf{-1};
        DevBuf d_cf;
        d_cf.alloc(8);
        HIP_CHECK(hipMemcpyAsync(d_cf.p, cf.data(), 4, hipMemcpyHostToDevice, stream));
        sp.sch.col_fixed = d_cf.as<int32_t>();
        sp.sch.n_static = 0;
        DevBuf d_sf;
        d_sf.alloc(8);
        sp.sch.static_fixed = d_sf.as<int32_t>();
        sp.sch.column_index_size = 64 * 1024;
        init_outstats(d_stats, stream);
        std::vector<std::pair<bytes, std::string>> cols = {
            {bytes{'v', 'a', 'l'}, "org.apache.cassandra.db.marshal.BytesType"}};
        write_sstable_device(opb, urows, n, sp, d_stats, d_tomb, tomb_cap, output_base,
                             "org.apache.cassandra.db.marshal.BytesType",
                             std::vector<std::string>{}, cols,
                             std::vector<std::pair<bytes, std::string>>{}, stream);
        HIP_CHECK(hipStreamDestroy(stream));
        return GPUC_OK;
    } catch (const std::exception& e) {
        set_err(error, error_len, e.what());
        return GPUC_ERR_INTERNAL;
    }
}


// ---- full-schema memtable flush (gpuc_flush_table) -----------------------
// Memtable.FlushablePartitionSet -> SortedTablePartitionWriter: the caller
// hands partitions with within-partition clustering order already
// established (memtables are sorted maps); the engine token-sorts the
// partitions on device, validates order/uniqueness, marshals rows into the
// same OutParts/UnfCols layout the generator and parse kernels produce, and
// runs the shared writer kernels (serialize/compress/index/bloom/stats).
namespace flushv2 {

// host mirrors of the device ck encodings (kernels_rows.hip ck_sortable /
// ck_prefix_var)
static uint64_t h_ck_sortable(const uint8_t* p, int width) {
    uint64_t v = 0;
    for (int b = 0; b < width; b++) v = (v << 8) | p[b];
    return v ^ (1ULL << (8 * width - 1));
}
static uint64_t h_ck_prefix_var(const uint8_t* p, uint32_t len) {
    uint64_t v = 0;
    uint32_t n = len < 8 ? len : 8;
    for (uint32_t b = 0; b < n; b++) v |= (uint64_t)p[b] << (8 * (7 - b));
    return v;
}
static int h_bk_cmp_to_clustering(uint8_t kk) {
    static const int tbl[8] = {-1, -1, -1, -1, 0, 1, 1, 1};
    return tbl[kk];
}
static int h_bk_comparison(uint8_t k) {
    static const int tbl[8] = {0, 0, 0, 1, 2, 3, 3, 3};
    return tbl[k];
}

// ClusteringComparator.compare over two ABI unfiltereds (mirrors pos_cmp)
static int h_unf_cmp(const gpuc_unfiltered& a, const gpuc_unfiltered& b,
                     const std::vector<int32_t>& ckw) {
    uint32_t mn = a.ck_count < b.ck_count ? a.ck_count : b.ck_count;
    for (uint32_t c = 0; c < mn; c++) {
        int32_t w = ckw[c];
        const uint8_t* pa = a.ck[c];
        const uint8_t* pb = b.ck[c];
        uint32_t la = a.ck_lens[c], lb = b.ck_lens[c];
        uint64_t sa, sb;
        if (w > 0) { sa = h_ck_sortable(pa, w); sb = h_ck_sortable(pb, w); }
        else { sa = h_ck_prefix_var(pa, la); sb = h_ck_prefix_var(pb, lb); }
        if (sa != sb) return sa < sb ? -1 : 1;
        if (w < 0) {
            uint32_t n = la < lb ? la : lb;
            for (uint32_t i = 8; i < n; i++)
                if (pa[i] != pb[i]) return pa[i] < pb[i] ? -1 : 1;
            if (la != lb) return la < lb ? -1 : 1;
        }
    }
    if (a.ck_count != b.ck_count)
        return a.ck_count < b.ck_count ? h_bk_cmp_to_clustering(a.kind)
                                       : -h_bk_cmp_to_clustering(b.kind);
    return h_bk_comparison(a.kind) - h_bk_comparison(b.kind);
}

}  // namespace flushv2

extern "C" int gpuc_flush_table(const gpuc_flush_schema* schema, const gpuc_flush_part* parts,
                                uint64_t n_parts, const char* output_base, int32_t device,
                                char* error, size_t error_len) {
    using namespace flushv2;
    try {
        if (gpuc_device_count() <= 0) { set_err(error, error_len, "no HIP device"); return GPUC_ERR_NO_GPU; }
        if (!schema || !parts || n_parts == 0) throw std::runtime_error("no partitions");
        const gpuc_flush_schema& S = *schema;
        if (S.n_cols == 0 || S.n_cols + S.n_static > 63)
            throw std::runtime_error("1..63 columns supported");
        if (S.n_cpx > 1) throw std::runtime_error("at most one complex column");
        if (S.n_ck > 32) throw std::runtime_error("at most 32 clustering columns");
        const std::string MAPT_PFX = "org.apache.cassandra.db.marshal.MapType";
        const std::string CTRT = "org.apache.cassandra.db.marshal.CounterColumnType";
        std::string key_type = S.key_type ? S.key_type
                                          : "org.apache.cassandra.db.marshal.BytesType";
        std::vector<std::string> ck_types;
        std::vector<int32_t> ckw_h;
        for (uint32_t i = 0; i < S.n_ck; i++) {
            ck_types.emplace_back(S.ck_types[i]);
            ckw_h.push_back(ck_type_width(ck_types.back()));
        }
        // n_cols in the ABI counts ALL regular columns; SchemaParams.n_cols
        // counts the SIMPLE ones (complex last, excluded)
        const uint32_t NSIMPLE = S.n_cols - S.n_cpx;
        std::vector<std::pair<bytes, std::string>> regular_cols, static_cols;
        std::vector<int32_t> colw_h, staticw_h;
        bool counters = false;
        for (uint32_t i = 0; i < S.n_cols; i++) {
            std::string t(S.col_types[i]);
            bool is_map = t.compare(0, MAPT_PFX.size(), MAPT_PFX) == 0;
            if (is_map != (S.n_cpx == 1 && i == S.n_cols - 1))
                throw std::runtime_error("a complex (MapType) column must be the last regular column");
            if (t == CTRT) counters = true;
            regular_cols.emplace_back(bytes(S.col_names[i], S.col_names[i] + S.col_name_lens[i]), t);
            if (i < NSIMPLE)
                colw_h.push_back(t == "org.apache.cassandra.db.marshal.LongType" ? 8
                                 : t == "org.apache.cassandra.db.marshal.Int32Type" ? 4 : -1);
        }
        for (uint32_t i = 0; i < S.n_static; i++) {
            std::string t(S.static_types[i]);
            static_cols.emplace_back(bytes(S.static_names[i], S.static_names[i] + S.static_name_lens[i]), t);
            staticw_h.push_back(t == "org.apache.cassandra.db.marshal.LongType" ? 8
                                : t == "org.apache.cassandra.db.marshal.Int32Type" ? 4 : -1);
        }
        // ---- pass 0: totals + validation ----
        uint64_t key_bytes = 0, n_unf = 0, val_bytes = 0, cpx_total = 0;
        for (uint64_t p = 0; p < n_parts; p++) {
            const gpuc_flush_part& P = parts[p];
            if (P.key_len == 0) throw std::runtime_error("empty partition key");
            key_bytes += P.key_len;
            n_unf += P.n_unf;
            if (P.static_flags && S.n_static)
                for (uint32_t c = 0; c < S.n_static; c++)
                    if (P.static_cells[c].flags & GPUC_CELLF_PRESENT)
                        val_bytes += P.static_cells[c].value_len;
            const gpuc_unfiltered* prev = nullptr;
            for (uint64_t r = 0; r < P.n_unf; r++) {
                const gpuc_unfiltered& U = P.unf[r];
                if (U.kind > 7 || U.kind == 3) throw std::runtime_error("bad unfiltered kind");
                bool is_row = U.kind == 4;
                if (is_row && U.ck_count != S.n_ck)
                    throw std::runtime_error("rows must have all clustering components");
                if (U.ck_count > S.n_ck) throw std::runtime_error("ck_count exceeds schema");
                for (uint32_t c = 0; c < U.ck_count; c++) {
                    int32_t w = ckw_h[c];
                    if (w > 0 && U.ck_lens[c] != (uint32_t)w)
                        throw std::runtime_error("fixed clustering component width mismatch");
                    if (w < 0) val_bytes += U.ck_lens[c];
                }
                if (is_row) {
                    for (uint32_t c = 0; c < NSIMPLE; c++)
                        if (U.cells[c].flags & GPUC_CELLF_PRESENT) val_bytes += U.cells[c].value_len;
                    if (U.has_cpx && !S.n_cpx)
                        throw std::runtime_error("complex data on a schema without a complex column");
                    if (U.has_cpx) {
                        cpx_total += U.n_cpx_cells;
                        for (uint32_t e = 0; e < U.n_cpx_cells; e++)
                            val_bytes += U.cpx_cells[e].path_len + U.cpx_cells[e].cell.value_len;
                    }
                }
                if (prev && h_unf_cmp(*prev, U, ckw_h) >= 0)
                    throw std::runtime_error("unfiltereds out of clustering order in partition " +
                                             std::to_string(p));
                prev = &U;
            }
        }
        HIP_CHECK(hipSetDevice(device));
        hipStream_t stream;
        HIP_CHECK(hipStreamCreate(&stream));
        ensure_crc_tables(stream);
        // ---- key arena + device partition sort (token, key bytes) ----
        std::vector<uint8_t> kbuf(key_bytes);
        std::vector<uint64_t> koff(n_parts + 1, 0);
        for (uint64_t p = 0; p < n_parts; p++) {
            memcpy(kbuf.data() + koff[p], parts[p].key, parts[p].key_len);
            koff[p + 1] = koff[p] + parts[p].key_len;
        }
        DevBuf d_keys, d_koff, d_a, d_b, d_koff_adj, d_err;
        d_keys.alloc(kbuf.size());
        d_koff.alloc((n_parts + 1) * 8);
        HIP_CHECK(hipMemcpyAsync(d_keys.p, kbuf.data(), kbuf.size(), hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(d_koff.p, koff.data(), (n_parts + 1) * 8, hipMemcpyHostToDevice, stream));
        FlushParams fp{};
        fp.keys = d_keys.as<uint8_t>();
        fp.key_off = d_koff.as<uint64_t>();
        fp.n = n_parts;
        d_a.alloc(n_parts * sizeof(MRec));
        d_b.alloc(n_parts * sizeof(MRec));
        uint32_t pblocks = (uint32_t)((n_parts + 255) / 256);
        hipLaunchKernelGGL(k_flush_recs, dim3(pblocks), dim3(256), 0, stream, fp, d_a.as<MRec>());
        std::vector<uint64_t> koff_adj(n_parts);
        for (uint64_t p = 0; p < n_parts; p++) koff_adj[p] = koff[p] - 2;
        d_koff_adj.alloc(n_parts * 8);
        HIP_CHECK(hipMemcpyAsync(d_koff_adj.p, koff_adj.data(), n_parts * 8, hipMemcpyHostToDevice, stream));
        KeyLut lut{};
        lut.base[0] = d_keys.as<uint8_t>();
        lut.pos[0] = d_koff_adj.as<uint64_t>();
        lut.enabled = 1;
        MRec* d_sorted = merge_sort_recs(d_a.as<MRec>(), d_b.as<MRec>(), n_parts, stream, lut);
        d_err.alloc(8);
        HIP_CHECK(hipMemsetAsync(d_err.p, 0, 8, stream));
        hipLaunchKernelGGL(k_flush_dupcheck, dim3(pblocks), dim3(256), 0, stream, fp, d_sorted,
                           d_err.as<unsigned long long>());
        std::vector<MRec> sorted(n_parts);
        HIP_CHECK(hipStreamSynchronize(stream));
        HIP_CHECK(hipMemcpy(sorted.data(), d_sorted, n_parts * sizeof(MRec), hipMemcpyDeviceToHost));
        {
            unsigned long long e = 0;
            HIP_CHECK(hipMemcpy(&e, d_err.p, 8, hipMemcpyDeviceToHost));
            if (e) throw std::runtime_error("duplicate partition keys in flush input");
        }
        // ---- device layout + host mirrors (filled in sorted order) ----
        OutPartsBuf opb;
        opb.alloc(n_parts, S.n_static);
        UnfColsBuf urows;
        urows.alloc(n_unf, NSIMPLE ? NSIMPLE : 1, S.n_ck, S.n_cpx);
        if (S.n_cpx) urows.alloc_cpx_arena(cpx_total);
        DevBuf d_vals;
        d_vals.alloc(val_bytes ? val_bytes : 1);
        uint8_t* const VBASE = d_vals.as<uint8_t>();
        std::vector<uint8_t> h_vals(val_bytes ? val_bytes : 1);
        uint64_t vcur = 0;
        auto stash = [&](const uint8_t* src, uint32_t len) -> uint64_t {
            uint64_t at = (uint64_t)(VBASE + vcur);
            if (len) memcpy(h_vals.data() + vcur, src, len);
            vcur += len;
            return at;
        };
        const uint32_t NCK = S.n_ck ? S.n_ck : 1;
        const uint32_t NCV = NSIMPLE ? NSIMPLE : 1;
        const uint32_t NST = S.n_static ? S.n_static : 1;
        // OutParts host
        std::vector<uint64_t> h_keypfx(n_parts), h_keyaddr(n_parts), h_rowbase(n_parts);
        std::vector<int64_t> h_token(n_parts), h_pdelm(n_parts);
        std::vector<uint16_t> h_klen(n_parts);
        std::vector<uint32_t> h_pdell(n_parts), h_rowcnt(n_parts);
        std::vector<uint8_t> h_keep(n_parts, 1);
        // statics host
        std::vector<uint8_t> h_sflags(n_parts, 0), h_scf(n_parts * NST, 0);
        std::vector<int64_t> h_sts(n_parts, NO_TIMESTAMP), h_slet(n_parts, NO_DELETION_TIME),
            h_sdm(n_parts, INT64_MIN), h_scts(n_parts * NST, 0);
        std::vector<int32_t> h_sttl(n_parts, 0), h_scttl(n_parts * NST, 0);
        std::vector<uint32_t> h_sdl(n_parts, LDT_NONE_U32), h_scldt(n_parts * NST, 0),
            h_scvl(n_parts * NST, 0);
        std::vector<uint64_t> h_scva(n_parts * NST, 0);
        // UnfCols host
        std::vector<uint64_t> h_ck(n_unf * NCK, 0), h_ckaddr(n_unf * NCK, 0);
        std::vector<uint32_t> h_cklen(n_unf * NCK, 0);
        std::vector<uint8_t> h_ckcnt(n_unf, 0), h_rkind(n_unf), h_flags(n_unf, 0),
            h_cf(n_unf * NCV, 0);
        std::vector<int64_t> h_lts(n_unf, NO_TIMESTAMP), h_llet(n_unf, NO_DELETION_TIME),
            h_rdm(n_unf, INT64_MIN), h_sm(n_unf, INT64_MIN), h_cts(n_unf * NCV, 0);
        std::vector<int32_t> h_lttl(n_unf, 0), h_cttl(n_unf * NCV, 0);
        std::vector<uint32_t> h_rdl(n_unf, LDT_NONE_U32), h_sl(n_unf, LDT_NONE_U32),
            h_cldt(n_unf * NCV, 0), h_cvl(n_unf * NCV, 0);
        std::vector<uint64_t> h_cva(n_unf * NCV, 0);
        std::vector<int64_t> h_xdm(S.n_cpx ? n_unf : 0);
        std::vector<uint32_t> h_xdl(S.n_cpx ? n_unf : 0), h_xcnt(S.n_cpx ? n_unf : 0);
        std::vector<uint64_t> h_xstart(S.n_cpx ? n_unf : 0);
        std::vector<int64_t> h_xts(cpx_total);
        std::vector<uint32_t> h_xldt(cpx_total), h_xpl(cpx_total), h_xvl(cpx_total);
        std::vector<int32_t> h_xttl(cpx_total);
        std::vector<uint8_t> h_xf(cpx_total);
        std::vector<uint64_t> h_xpa(cpx_total), h_xva(cpx_total);
        auto conv_cell = [&](const gpuc_cell& C, int64_t* ts, uint32_t* ldt, int32_t* ttl,
                             uint64_t* va, uint32_t* vl, uint8_t* cf) {
            if (!(C.flags & GPUC_CELLF_PRESENT)) { *cf = 0; return; }
            uint8_t f = CELLF_PRESENT;
            if (C.flags & GPUC_CELLF_HAS_VALUE) f |= CELLF_HAS_VALUE;
            if (C.ttl != 0) f |= CELLF_EXPIRING;
            *cf = f;
            *ts = C.ts;
            *ldt = C.ldt;
            *ttl = C.ttl;
            *vl = C.value_len;
            *va = (C.flags & GPUC_CELLF_HAS_VALUE) ? stash(C.value, C.value_len)
                                                   : (uint64_t)VBASE;
        };
        uint64_t ob = 0, xcur = 0;
        for (uint64_t si = 0; si < n_parts; si++) {
            const uint64_t p = sorted[si].idx;
            const gpuc_flush_part& P = parts[p];
            h_keypfx[si] = sorted[si].pfx;
            h_token[si] = (int64_t)(sorted[si].tok ^ 0x8000000000000000ULL);
            h_keyaddr[si] = (uint64_t)(d_keys.as<uint8_t>() + koff[p]);
            h_klen[si] = P.key_len;
            h_pdelm[si] = P.pdel_mfda;
            h_pdell[si] = P.pdel_ldt;
            h_rowbase[si] = ob;
            h_rowcnt[si] = (uint32_t)P.n_unf;
            if (S.n_static && P.static_flags) {
                h_sflags[si] = P.static_flags & 7;  // PF_HAS_ROW|PF_LIVE_TS|PF_ROW_DEL
                h_sts[si] = P.static_live_ts;
                h_sttl[si] = P.static_live_ttl;
                h_slet[si] = P.static_live_let;
                h_sdm[si] = P.static_del_mfda;
                h_sdl[si] = P.static_del_ldt;
                for (uint32_t c = 0; c < S.n_static; c++)
                    conv_cell(P.static_cells[c], &h_scts[si * NST + c], &h_scldt[si * NST + c],
                              &h_scttl[si * NST + c], &h_scva[si * NST + c], &h_scvl[si * NST + c],
                              &h_scf[si * NST + c]);
            }
            for (uint64_t r = 0; r < P.n_unf; r++, ob++) {
                const gpuc_unfiltered& U = P.unf[r];
                h_rkind[ob] = U.kind;
                h_ckcnt[ob] = U.ck_count;
                for (uint32_t c = 0; c < U.ck_count; c++) {
                    uint64_t oc = ob * NCK + c;
                    int32_t w = ckw_h[c];
                    if (w > 0) {
                        h_ck[oc] = h_ck_sortable(U.ck[c], w);
                        h_ckaddr[oc] = 0;
                        h_cklen[oc] = (uint32_t)w;
                    } else {
                        h_ck[oc] = h_ck_prefix_var(U.ck[c], U.ck_lens[c]);
                        h_ckaddr[oc] = stash(U.ck[c], U.ck_lens[c]);
                        h_cklen[oc] = U.ck_lens[c];
                    }
                }
                if (U.kind == 4) {
                    uint8_t f = PF_HAS_ROW;
                    if (U.row_flags & GPUC_ROWF_LIVE_TS) {
                        f |= PF_LIVE_TS;
                        h_lts[ob] = U.live_ts;
                        h_lttl[ob] = U.live_ttl;
                        h_llet[ob] = U.live_let;
                    }
                    if (U.row_flags & GPUC_ROWF_DELETED) {
                        f |= PF_ROW_DEL;
                        h_rdm[ob] = U.del_mfda;
                        h_rdl[ob] = U.del_ldt;
                    }
                    for (uint32_t c = 0; c < NSIMPLE; c++)
                        conv_cell(U.cells[c], &h_cts[ob * NCV + c], &h_cldt[ob * NCV + c],
                                  &h_cttl[ob * NCV + c], &h_cva[ob * NCV + c],
                                  &h_cvl[ob * NCV + c], &h_cf[ob * NCV + c]);
                    if (S.n_cpx) {
                        h_xdm[ob] = INT64_MIN;
                        h_xdl[ob] = LDT_NONE_U32;
                        h_xstart[ob] = xcur;
                        h_xcnt[ob] = 0;
                        if (U.has_cpx) {
                            f |= PF_HAS_CPX;
                            h_xdm[ob] = U.cpx_del_mfda;
                            h_xdl[ob] = U.cpx_del_ldt;
                            h_xcnt[ob] = U.n_cpx_cells;
                            for (uint32_t e = 0; e < U.n_cpx_cells; e++, xcur++) {
                                const gpuc_cpx_cell& X = U.cpx_cells[e];
                                conv_cell(X.cell, &h_xts[xcur], &h_xldt[xcur], &h_xttl[xcur],
                                          &h_xva[xcur], &h_xvl[xcur], &h_xf[xcur]);
                                h_xpa[xcur] = stash(X.path, X.path_len);
                                h_xpl[xcur] = X.path_len;
                            }
                        }
                    }
                    h_flags[ob] = f;
                } else {
                    h_flags[ob] = 0;
                    h_rdm[ob] = U.del_mfda;
                    h_rdl[ob] = U.del_ldt;
                    if (U.kind == 2 || U.kind == 5) {  // boundary: open deletion
                        h_sm[ob] = U.open_mfda;
                        h_sl[ob] = U.open_ldt;
                    }
                    if (S.n_cpx) {
                        h_xdm[ob] = INT64_MIN;
                        h_xdl[ob] = LDT_NONE_U32;
                        h_xstart[ob] = xcur;
                        h_xcnt[ob] = 0;
                    }
                }
            }
        }
        // ---- upload ----
        auto up = [&](const DevBuf& d, const void* src, size_t len) {
            if (len) HIP_CHECK(hipMemcpyAsync(d.p, src, len, hipMemcpyHostToDevice, stream));
        };
        up(d_vals, h_vals.data(), vcur);
        up(opb.keypfx, h_keypfx.data(), n_parts * 8);
        up(opb.key_addr, h_keyaddr.data(), n_parts * 8);
        up(opb.token, h_token.data(), n_parts * 8);
        up(opb.klen, h_klen.data(), n_parts * 2);
        up(opb.pdel_mfda, h_pdelm.data(), n_parts * 8);
        up(opb.pdel_ldt, h_pdell.data(), n_parts * 4);
        up(opb.row_base, h_rowbase.data(), n_parts * 8);
        up(opb.row_count, h_rowcnt.data(), n_parts * 4);
        up(opb.keep, h_keep.data(), n_parts);
        if (S.n_static) {
            up(opb.stb.flags, h_sflags.data(), n_parts);
            up(opb.stb.live_ts, h_sts.data(), n_parts * 8);
            up(opb.stb.live_ttl, h_sttl.data(), n_parts * 4);
            up(opb.stb.live_let, h_slet.data(), n_parts * 8);
            up(opb.stb.rdel_mfda, h_sdm.data(), n_parts * 8);
            up(opb.stb.rdel_ldt, h_sdl.data(), n_parts * 4);
            up(opb.stb.cell_flags, h_scf.data(), n_parts * NST);
            up(opb.stb.cell_ts, h_scts.data(), n_parts * NST * 8);
            up(opb.stb.cell_ldt, h_scldt.data(), n_parts * NST * 4);
            up(opb.stb.cell_ttl, h_scttl.data(), n_parts * NST * 4);
            up(opb.stb.val_addr, h_scva.data(), n_parts * NST * 8);
            up(opb.stb.val_len, h_scvl.data(), n_parts * NST * 4);
        }
        up(urows.ck, h_ck.data(), n_unf * NCK * 8);
        up(urows.ck_addr, h_ckaddr.data(), n_unf * NCK * 8);
        up(urows.ck_len, h_cklen.data(), n_unf * NCK * 4);
        up(urows.ck_count, h_ckcnt.data(), n_unf);
        up(urows.rkind, h_rkind.data(), n_unf);
        up(urows.flags, h_flags.data(), n_unf);
        up(urows.live_ts, h_lts.data(), n_unf * 8);
        up(urows.live_ttl, h_lttl.data(), n_unf * 4);
        up(urows.live_let, h_llet.data(), n_unf * 8);
        up(urows.rdel_mfda, h_rdm.data(), n_unf * 8);
        up(urows.rdel_ldt, h_rdl.data(), n_unf * 4);
        up(urows.start_mfda, h_sm.data(), n_unf * 8);
        up(urows.start_ldt, h_sl.data(), n_unf * 4);
        up(urows.cell_ts, h_cts.data(), n_unf * NCV * 8);
        up(urows.cell_ldt, h_cldt.data(), n_unf * NCV * 4);
        up(urows.cell_ttl, h_cttl.data(), n_unf * NCV * 4);
        up(urows.val_addr, h_cva.data(), n_unf * NCV * 8);
        up(urows.val_len, h_cvl.data(), n_unf * NCV * 4);
        up(urows.cell_flags, h_cf.data(), n_unf * NCV);
        if (S.n_cpx) {
            up(urows.cpx_del_mfda, h_xdm.data(), n_unf * 8);
            up(urows.cpx_del_ldt, h_xdl.data(), n_unf * 4);
            up(urows.cpx_start, h_xstart.data(), n_unf * 8);
            up(urows.cpx_count, h_xcnt.data(), n_unf * 4);
            if (cpx_total) {
                up(urows.cx_ts, h_xts.data(), cpx_total * 8);
                up(urows.cx_ldt, h_xldt.data(), cpx_total * 4);
                up(urows.cx_ttl, h_xttl.data(), cpx_total * 4);
                up(urows.cx_flags, h_xf.data(), cpx_total);
                up(urows.cx_pa, h_xpa.data(), cpx_total * 8);
                up(urows.cx_pl, h_xpl.data(), cpx_total * 4);
                up(urows.cx_va, h_xva.data(), cpx_total * 8);
                up(urows.cx_vl, h_xvl.data(), cpx_total * 4);
            }
        }
        // ---- header mins from a pre-collect (gpuc_generate's recipe) ----
        DevBuf d_stats, d_tomb;
        d_stats.alloc(sizeof(OutStats));
        init_outstats(d_stats, stream);
        uint32_t tomb_cap = (uint32_t)std::min<uint64_t>(
            n_unf * ((uint64_t)NSIMPLE + S.n_cpx + 2) + cpx_total +
                n_parts * ((uint64_t)S.n_static + 2) + 1024,
            400000000ull);
        d_tomb.alloc((uint64_t)tomb_cap * 4);
        uint32_t blocks = (uint32_t)((n_parts + 255) / 256);
        hipLaunchKernelGGL(k_collect_rows, dim3(blocks), dim3(256), 0, stream, opb.op, urows.uc,
                           n_parts, NCV, d_stats.as<OutStats>(), d_tomb.as<uint32_t>(), tomb_cap);
        OutStats hs0;
        HIP_CHECK(hipStreamSynchronize(stream));
        HIP_CHECK(hipMemcpy(&hs0, d_stats.p, sizeof(OutStats), hipMemcpyDeviceToHost));
        SerParams2 sp{};
        if (S.has_stats) {
            // caller-provided EncodingStats (Memtable -> SerializationHeader.make)
            sp.hs.min_ts = S.stats_min_ts;
            sp.hs.min_ldt = S.stats_min_ldt;
            sp.hs.min_ttl = S.stats_min_ttl;
        } else {
            sp.hs.min_ts = hs0.min_ts_flip == 0xFFFFFFFFFFFFFFFFULL
                               ? TIMESTAMP_EPOCH : (int64_t)(hs0.min_ts_flip ^ 0x8000000000000000ULL);
            sp.hs.min_ldt = hs0.min_ldt_flip == 0xFFFFFFFFFFFFFFFFULL
                                ? DELETION_TIME_EPOCH : (int64_t)(hs0.min_ldt_flip ^ 0x8000000000000000ULL);
            if (sp.hs.min_ldt == NO_DELETION_TIME) sp.hs.min_ldt = DELETION_TIME_EPOCH;
            sp.hs.min_ttl = hs0.min_ttl == 0xFFFFFFFFu ? 0 : (int32_t)hs0.min_ttl;
        }
        sp.sch.n_ck = S.n_ck;
        DevBuf d_ckw, d_cf2, d_sf2;
        d_ckw.alloc(ckw_h.size() * 4 + 8);
        if (S.n_ck) up(d_ckw, ckw_h.data(), ckw_h.size() * 4);
        sp.sch.ck_w = d_ckw.as<int32_t>();
        sp.sch.n_cols = NSIMPLE;
        d_cf2.alloc(colw_h.size() * 4 + 8);
        if (NSIMPLE) up(d_cf2, colw_h.data(), colw_h.size() * 4);
        sp.sch.col_fixed = d_cf2.as<int32_t>();
        sp.sch.n_static = S.n_static;
        d_sf2.alloc(staticw_h.size() * 4 + 8);
        if (S.n_static) up(d_sf2, staticw_h.data(), staticw_h.size() * 4);
        sp.sch.static_fixed = d_sf2.as<int32_t>();
        sp.sch.n_cpx = S.n_cpx;
        sp.sch.counters = counters ? 1u : 0u;
        sp.sch.column_index_size = S.column_index_size ? S.column_index_size : 64 * 1024;
        init_outstats(d_stats, stream);
        write_sstable_device(opb, urows, n_parts, sp, d_stats, d_tomb, tomb_cap, output_base,
                             key_type, ck_types, regular_cols, static_cols, stream, 0,
                             S.snappy != 0, S.bti != 0);
        HIP_CHECK(hipStreamDestroy(stream));
        return GPUC_OK;
    } catch (const std::exception& e) {
        set_err(error, error_len, e.what());
        return GPUC_ERR_INTERNAL;
    }
}

extern "C" int gpuc_validate(const gpuc_job* job, const char* out_path, uint64_t* n_partitions,
                             char* error, size_t error_len) {
    if (job->n_output_shards > 1 || job->n_tomb_sources > 0 || job->n_keep_ranges > 0) {
        set_err(error, error_len, "validation: sharding/tomb-source/keep-range modes unsupported");
        return GPUC_ERR_UNSUPPORTED;
    }
    gpuc_result res{};
    g_validate_out = out_path;
    g_validate_count = 0;
    int rc = gpuc_compact(job, &res);
    g_validate_out = nullptr;
    if (rc == GPUC_OK && n_partitions) *n_partitions = g_validate_count;
    if (rc != GPUC_OK) set_err(error, error_len, res.error);
    return rc;
}

extern "C" int gpuc_generate(const gpuc_gen_spec* spec, const char* dir, char* error, size_t error_len) {
    try {
        if (gpuc_device_count() <= 0) { set_err(error, error_len, "no HIP device"); return GPUC_ERR_NO_GPU; }
        HIP_CHECK(hipSetDevice(spec->device));
        hipStream_t stream;
        HIP_CHECK(hipStreamCreate(&stream));
        ensure_crc_tables(stream);
        uint64_t R = spec->rows_per_sstable;
        uint64_t stride = R * (100 - spec->overlap_pct) / 100;
        uint64_t universe = std::max<uint64_t>(stride * spec->n_sstables, R);
        for (uint32_t s = 0; s < spec->n_sstables; s++) {
            GenParams2 gp{};
            gp.seed = spec->seed;
            gp.universe = universe;
            gp.stride = stride;
            gp.rows = R;
            gp.sst = s;
            gp.value_len = spec->value_len;
            gp.value_repeat_pct = spec->value_repeat_pct;
            gp.tombstone_pct = spec->tombstone_pct;
            gp.partition_del_pct = spec->partition_del_pct;
            gp.clustering_rows = spec->clustering_rows;
            gp.range_tomb_pct = spec->range_tomb_pct;
            gp.base_ts = spec->base_ts;
            gp.base_ldt = spec->base_ldt;
            gp.key_len = spec->key_len ? spec->key_len : 8;
            gp.ck_text = spec->ck_text;
            gp.n_value_cols = spec->n_value_cols ? spec->n_value_cols : 1;
            gp.ck_cols = spec->ck_cols;
            gp.static_pct = spec->static_pct;
            if (gp.ck_cols > 2) throw std::runtime_error("ck_cols must be 0..2");
            if (gp.ck_cols == 2 && gp.ck_text) throw std::runtime_error("ck_cols=2 with ck_text unsupported");
            gp.col_missing_pct = spec->col_missing_pct;
            gp.ttl_pct = spec->ttl_pct;
            gp.complex_pct = spec->complex_pct;
            gp.complex_del_pct = spec->complex_del_pct;
            gp.counter = spec->counter;
            if (gp.counter) {
                if (gp.complex_pct || spec->static_pct || spec->ttl_pct || gp.n_value_cols > 1)
                    throw std::runtime_error("counter mode: single counter column only");
                gp.value_len = 274;  // context arena stride (8-shard worst case)
                // sorted CounterId pool (oracle gen.h gen_counter_id)
                struct P { uint8_t id[16]; uint8_t idx; };
                std::vector<P> pool(8);
                for (uint32_t i = 0; i < 8; i++) {
                    uint64_t a = splitmix64(0xC0C0C0C0ULL + i), b = splitmix64(0xF00DF00DULL + i);
                    for (int x = 0; x < 8; x++) pool[i].id[x] = (uint8_t)(a >> (8 * (7 - x)));
                    for (int x = 0; x < 8; x++) pool[i].id[8 + x] = (uint8_t)(b >> (8 * (7 - x)));
                    pool[i].idx = (uint8_t)i;
                }
                std::sort(pool.begin(), pool.end(), [](const P& x, const P& y) {
                    return memcmp(x.id, y.id, 16) < 0;
                });
                for (int i = 0; i < 8; i++) {
                    memcpy(gp.ctr_pool[i], pool[i].id, 16);
                    gp.ctr_pool_idx[i] = pool[i].idx;
                }
            }
            if (gp.n_value_cols > 63) throw std::runtime_error("n_value_cols must be 1..63");
            if (gp.ck_text && (uint64_t)gp.clustering_rows * 16 >= 100000000ull)
                throw std::runtime_error("ck_text needs clustering_rows*16 < 1e8 (8-digit order)");
            if (gp.key_len < 8 || gp.key_len > 255)
                throw std::runtime_error("key_len must be 8..255 (generator contract)");
            KeyLut lut{};  // generated 8-byte prefixes are unique: prefix compare is exact
            DevBuf d_a, d_b, d_ids, d_keys, d_vals, d_stats, d_tomb, d_prows;
            d_a.alloc(R * sizeof(MRec));
            d_b.alloc(R * sizeof(MRec));
            d_ids.alloc(R * 8);
            d_keys.alloc(R * (uint64_t)gp.key_len);
            uint32_t blocks = (uint32_t)((R + 255) / 256);
            hipLaunchKernelGGL(k_gen_recs2, dim3(blocks), dim3(256), 0, stream, gp,
                               d_a.as<MRec>(), d_ids.as<uint64_t>(), d_keys.as<uint8_t>());
            MRec* d_sorted = merge_sort_recs(d_a.as<MRec>(), d_b.as<MRec>(), R, stream, lut);
            OutPartsBuf opb;
            opb.alloc(R, gp.static_pct ? 1 : 0);
            d_prows.alloc(R * 8);
            DevBuf d_cpxcnt, d_cpxbytes;
            if (gp.complex_pct) d_cpxcnt.alloc(R * 8 + 8);
            hipLaunchKernelGGL(k_gen_count, dim3(blocks), dim3(256), 0, stream, gp, d_sorted,
                               d_ids.as<uint64_t>(), R, opb.op, d_prows.as<uint64_t>(),
                               d_keys.as<uint8_t>(),
                               gp.complex_pct ? d_cpxcnt.as<uint64_t>() : nullptr);
            uint64_t total_cpx = gp.complex_pct ? exscan_u64(d_cpxcnt.as<uint64_t>(), R, stream) : 0;
            uint64_t total_rows = exscan_u64(d_prows.as<uint64_t>(), R, stream);
            UnfColsBuf rows;
            uint32_t gen_nck = gp.clustering_rows ? (gp.ck_cols ? gp.ck_cols : 1) : 0;
            rows.alloc(total_rows, gp.n_value_cols, gen_nck, gp.complex_pct ? 1 : 0);
            if (gp.complex_pct) {
                rows.alloc_cpx_arena(total_cpx);
                d_cpxbytes.alloc(total_cpx * 12 + 16);  // 4B path + 8B value per cell
            }
            d_vals.alloc(total_rows * (uint64_t)gp.n_value_cols * spec->value_len);
            DevBuf d_ckarena, d_svals;
            if (gp.ck_text) d_ckarena.alloc(total_rows * 16 + 16);
            if (gp.static_pct) d_svals.alloc(R * (uint64_t)spec->value_len + 16);
            hipLaunchKernelGGL(k_gen_fill2, dim3(blocks), dim3(256), 0, stream, gp, d_sorted,
                               d_ids.as<uint64_t>(), R, opb.op, rows.uc, d_prows.as<uint64_t>(),
                               d_vals.as<uint8_t>(), gp.ck_text ? d_ckarena.as<uint8_t>() : nullptr,
                               gp.static_pct ? d_svals.as<uint8_t>() : nullptr,
                               gp.complex_pct ? d_cpxbytes.as<uint8_t>() : nullptr,
                               gp.complex_pct ? d_cpxcnt.as<uint64_t>() : nullptr);
            hipLaunchKernelGGL(k_gen_values2, dim3(blocks), dim3(256), 0, stream, gp, d_sorted,
                               d_ids.as<uint64_t>(), opb.op, rows.uc, R, d_vals.as<uint8_t>(),
                               d_prows.as<uint64_t>());
            d_stats.alloc(sizeof(OutStats));
            init_outstats(d_stats, stream);
            uint32_t tomb_cap = (uint32_t)std::min<uint64_t>(
                total_rows * ((uint64_t)gp.n_value_cols + (gp.complex_pct ? 7 : 0) + 2) +
                    R * 3 + 1024,
                400000000ull);
            d_tomb.alloc((uint64_t)tomb_cap * 4);
            {
                hipLaunchKernelGGL(k_collect_rows, dim3(blocks), dim3(256), 0, stream, opb.op,
                                   rows.uc, R, gp.n_value_cols, d_stats.as<OutStats>(),
                                   d_tomb.as<uint32_t>(), tomb_cap);
            }
            OutStats hs0;
            HIP_CHECK(hipStreamSynchronize(stream));
            HIP_CHECK(hipMemcpy(&hs0, d_stats.p, sizeof(OutStats), hipMemcpyDeviceToHost));
            SerParams2 sp{};
            sp.hs.min_ts = hs0.min_ts_flip == 0xFFFFFFFFFFFFFFFFULL
                               ? TIMESTAMP_EPOCH : (int64_t)(hs0.min_ts_flip ^ 0x8000000000000000ULL);
            sp.hs.min_ldt = hs0.min_ldt_flip == 0xFFFFFFFFFFFFFFFFULL
                                ? DELETION_TIME_EPOCH : (int64_t)(hs0.min_ldt_flip ^ 0x8000000000000000ULL);
            if (sp.hs.min_ldt == NO_DELETION_TIME) sp.hs.min_ldt = DELETION_TIME_EPOCH;
            sp.hs.min_ttl = hs0.min_ttl == 0xFFFFFFFFu ? 0 : (int32_t)hs0.min_ttl;
            sp.sch.n_ck = gen_nck;
            std::vector<int32_t> ckw_h(gen_nck, gp.ck_text ? -1 : 8);
            DevBuf d_ckw;
            d_ckw.alloc(ckw_h.size() * 4 + 8);
            if (gen_nck)
                HIP_CHECK(hipMemcpyAsync(d_ckw.p, ckw_h.data(), ckw_h.size() * 4,
                                         hipMemcpyHostToDevice, stream));
            sp.sch.ck_w = d_ckw.as<int32_t>();
            sp.sch.n_static = gp.static_pct ? 1 : 0;
            std::vector<int32_t> sfx_h(1, -1);
            DevBuf d_sfx;
            d_sfx.alloc(8);
            HIP_CHECK(hipMemcpyAsync(d_sfx.p, sfx_h.data(), 4, hipMemcpyHostToDevice, stream));
            sp.sch.static_fixed = d_sfx.as<int32_t>();
            sp.sch.n_cols = gp.n_value_cols;
            sp.sch.n_cpx = gp.complex_pct ? 1 : 0;
            std::vector<int32_t> gcf(gp.n_value_cols, -1);  // val blobs
            DevBuf d_gcf;
            d_gcf.alloc(gcf.size() * 4);
            HIP_CHECK(hipMemcpyAsync(d_gcf.p, gcf.data(), gcf.size() * 4,
                                     hipMemcpyHostToDevice, stream));
            sp.sch.col_fixed = d_gcf.as<int32_t>();
            sp.sch.column_index_size = 64 * 1024;
            // reset and recollect so the writer sees fresh stats (collect ran
            // once above only to derive the header mins)
            init_outstats(d_stats, stream);
            std::string base = spec->bti
                ? std::string(dir) + "/da-" + std::to_string(spec->first_generation + s) + "-bti"
                : std::string(dir) + "/oa-" + std::to_string(spec->first_generation + s) + "-big";
            std::vector<std::pair<bytes, std::string>> cols;
            if (gp.counter) {
                cols.push_back({bytes{'c', 'n', 't'}, "org.apache.cassandra.db.marshal.CounterColumnType"});
            } else if (gp.n_value_cols == 1) {
                cols.push_back({bytes{'v', 'a', 'l'}, "org.apache.cassandra.db.marshal.BytesType"});
            } else {
                for (uint32_t c = 0; c < gp.n_value_cols; c++) {
                    std::string nm = "val" + std::to_string(c);
                    cols.push_back({bytes(nm.begin(), nm.end()),
                                    "org.apache.cassandra.db.marshal.BytesType"});
                }
            }
            if (gp.complex_pct)
                cols.push_back({bytes{'z', 'm'}, "org.apache.cassandra.db.marshal.MapType(org.apache.cassandra.db.marshal.BytesType,org.apache.cassandra.db.marshal.BytesType)"});
            std::vector<std::pair<bytes, std::string>> scols;
            if (gp.static_pct)
                scols.push_back({bytes{'s', '0'}, "org.apache.cassandra.db.marshal.BytesType"});
            write_sstable_device(opb, rows, R, sp, d_stats, d_tomb, tomb_cap, base,
                                 gp.key_len > 8 ? "org.apache.cassandra.db.marshal.BytesType"
                                                : "org.apache.cassandra.db.marshal.LongType",
                                 std::vector<std::string>(
                                     gen_nck, gp.ck_text ? "org.apache.cassandra.db.marshal.UTF8Type"
                                                         : "org.apache.cassandra.db.marshal.LongType"),
                                 cols, scols, stream, 0, spec->snappy != 0, spec->bti != 0);
        }
        HIP_CHECK(hipStreamDestroy(stream));
        return GPUC_OK;
    } catch (const std::exception& e) {
        set_err(error, error_len, e.what());
        return GPUC_ERR_INTERNAL;
    }
}

