// ORACLE — test infrastructure only.
// CPU restatement of apache/cassandra byte-level codecs, used exclusively as
// the parity checker (tests/, __graft_entry__.smoke, bench.py cpu_baseline).
// The product path (cassandra_amd/csrc) must never link or call this.
#pragma once
#include <cstdint>
#include <cstring>
#include <cstdio>
#include <cstdlib>
#include <string>
#include <vector>
#include <stdexcept>

namespace oracle {

using bytes = std::vector<uint8_t>;

// ---------------------------------------------------------------------------
// Big-endian primitives (Java DataOutput semantics).
// ---------------------------------------------------------------------------
inline void put_be16(bytes& out, uint16_t v) { out.push_back(v >> 8); out.push_back(v); }
inline void put_be32(bytes& out, uint32_t v) { for (int i = 3; i >= 0; i--) out.push_back(v >> (8 * i)); }
inline void put_be64(bytes& out, uint64_t v) { for (int i = 7; i >= 0; i--) out.push_back((uint8_t)(v >> (8 * i))); }
inline void put_le32(bytes& out, uint32_t v) { for (int i = 0; i < 4; i++) out.push_back(v >> (8 * i)); }

struct Reader {
    const uint8_t* p;
    size_t len;
    size_t pos = 0;
    Reader(const uint8_t* p_, size_t n) : p(p_), len(n) {}
    explicit Reader(const bytes& b) : p(b.data()), len(b.size()) {}
    bool eof() const { return pos >= len; }
    size_t remaining() const { return len - pos; }
    void need(size_t n) const { if (pos + n > len) throw std::runtime_error("oracle.Reader: short read at " + std::to_string(pos)); }
    uint8_t u8() { need(1); return p[pos++]; }
    uint16_t be16() { need(2); uint16_t v = ((uint16_t)p[pos] << 8) | p[pos + 1]; pos += 2; return v; }
    uint32_t be32() { need(4); uint32_t v = 0; for (int i = 0; i < 4; i++) v = (v << 8) | p[pos + i]; pos += 4; return v; }
    uint64_t be64() { need(8); uint64_t v = 0; for (int i = 0; i < 8; i++) v = (v << 8) | p[pos + i]; pos += 8; return v; }
    uint32_t le32() { need(4); uint32_t v = 0; for (int i = 3; i >= 0; i--) v = (v << 8) | p[pos + i]; pos += 4; return v; }
    bytes take(size_t n) { need(n); bytes b(p + pos, p + pos + n); pos += n; return b; }
    void skip(size_t n) { need(n); pos += n; }
};

// ---------------------------------------------------------------------------
// Cassandra unsigned/signed vints — utils/vint/VIntCoding.java.
// First byte carries N leading 1-bits = number of EXTRA bytes; value packed
// big-endian in the remaining bits (readUnsignedVInt, VIntCoding.java:93-111;
// writeUnsignedVInt, :303-327). 9-byte case: 0xFF prefix + raw BE u64.
// ---------------------------------------------------------------------------
inline int unsigned_vint_size(uint64_t v) {
    // VIntCoding.computeUnsignedVIntSize (:535-540)
    int magnitude = __builtin_clzll(v | 1);
    return (639 - magnitude * 9) >> 6;
}
inline void put_unsigned_vint(bytes& out, uint64_t v) {
    int size = unsigned_vint_size(v);
    if (size == 1) { out.push_back((uint8_t)v); return; }
    if (size == 9) { out.push_back(0xFF); put_be64(out, v); return; }
    int extra = size - 1;
    uint8_t first_mask = (uint8_t)~(0xFFu >> extra);  // encodeExtraBytesToRead
    uint64_t reg = v << ((8 - size) * 8);
    out.push_back((uint8_t)((reg >> 56) | first_mask));
    for (int i = 1; i < size; i++) out.push_back((uint8_t)(reg >> (8 * (7 - i))));
}
inline uint64_t read_unsigned_vint(Reader& r) {
    int8_t first = (int8_t)r.u8();
    if (first >= 0) return (uint64_t)first;
    // numberOfExtraBytesToRead(firstByte) = nlz(~signextend(firstByte)) - 24
    uint32_t inv = ~(uint32_t)(int32_t)first;  // Java nlz(0) == 32
    int extra = inv == 0 ? 8 : __builtin_clz(inv) - 24;
    uint64_t v = (uint8_t)first & (0xFFu >> extra);
    for (int i = 0; i < extra; i++) v = (v << 8) | r.u8();
    return v;
}
inline uint64_t zigzag(int64_t n) { return ((uint64_t)n << 1) ^ (uint64_t)(n >> 63); }
inline int64_t unzigzag(uint64_t n) { return (int64_t)(n >> 1) ^ -(int64_t)(n & 1); }
inline void put_vint(bytes& out, int64_t v) { put_unsigned_vint(out, zigzag(v)); }
inline int64_t read_vint(Reader& r) { return unzigzag(read_unsigned_vint(r)); }
inline int vint_size(int64_t v) { return unsigned_vint_size(zigzag(v)); }

// Short length: unsigned 16-bit big-endian + raw bytes
// (utils/ByteBufferUtil.java writeWithShortLength/readWithShortLength).
inline void put_short_len_bytes(bytes& out, const bytes& b) {
    if (b.size() > 0xFFFF) throw std::runtime_error("short-length overflow");
    put_be16(out, (uint16_t)b.size());
    out.insert(out.end(), b.begin(), b.end());
}
inline void put_vint_len_bytes(bytes& out, const bytes& b) {
    // ByteBufferUtil.writeWithVIntLength
    put_unsigned_vint(out, b.size());
    out.insert(out.end(), b.begin(), b.end());
}
// Java DataOutput.writeUTF: u16 BE length + modified UTF-8 (our strings are ASCII).
inline void put_utf(bytes& out, const std::string& s) {
    put_be16(out, (uint16_t)s.size());
    out.insert(out.end(), s.begin(), s.end());
}

// ---------------------------------------------------------------------------
// MurmurHash.hash3_x64_128 — utils/MurmurHash.java:178-253.
// Cassandra's "*almost* MurmurHash 3.0" (MurmurHash.java:31): tail bytes are
// SIGN-EXTENDED (key.get() returns signed byte, no & 0xff) — kept bit-exactly.
// ---------------------------------------------------------------------------
inline uint64_t rotl64_(uint64_t v, int n) { return (v << n) | (v >> (64 - n)); }
inline uint64_t fmix_(uint64_t k) {
    k ^= k >> 33; k *= 0xff51afd7ed558ccdULL;
    k ^= k >> 33; k *= 0xc4ceb9fe1a85ec53ULL;
    k ^= k >> 33; return k;
}
// MurmurHash.hash2_64 (utils/MurmurHash.java:96-150) — the partition-key
// hash feeding the COMPACTION HyperLogLogPlus (MetadataCollector.java:180-183).
// NOTE the reference's tail bytes are SIGN-EXTENDED (java byte, no & 0xff):
// transcribed exactly, fixture-pinned (test_compaction_hll_fixture_pin).
inline uint64_t murmur2_64_cassandra(const uint8_t* key, size_t length, uint64_t seed) {
    const uint64_t m = 0xc6a4a7935bd1e995ULL;
    const int r = 47;
    uint64_t h = (seed & 0xffffffffULL) ^ (m * (uint64_t)length);
    size_t nl = length >> 3;
    for (size_t i = 0; i < nl; i++) {
        uint64_t k;
        memcpy(&k, key + i * 8, 8);  // little-endian load
        k *= m;
        k ^= k >> r;
        k *= m;
        h ^= k;
        h *= m;
    }
    size_t rem = length & 7;
    if (rem) {
        const uint8_t* t = key + length - rem;
        for (size_t b = rem; b-- > 0;)
            h ^= (uint64_t)(int64_t)(int8_t)t[b] << (8 * b);
        h *= m;
    }
    h ^= h >> r;
    h *= m;
    h ^= h >> r;
    return h;
}

inline void murmur3_128_cassandra(const uint8_t* key, size_t length, uint64_t seed,
                                  uint64_t out[2]) {
    const size_t nblocks = length >> 4;
    uint64_t h1 = seed, h2 = seed;
    const uint64_t c1 = 0x87c37b91114253d5ULL, c2 = 0x4cf5ad432745937fULL;
    for (size_t i = 0; i < nblocks; i++) {
        uint64_t k1, k2;  // getBlock: little-endian u64 with & 0xff (MurmurHash.java:152-160)
        memcpy(&k1, key + i * 16, 8);
        memcpy(&k2, key + i * 16 + 8, 8);
        k1 *= c1; k1 = rotl64_(k1, 31); k1 *= c2; h1 ^= k1;
        h1 = rotl64_(h1, 27); h1 += h2; h1 = h1 * 5 + 0x52dce729;
        k2 *= c2; k2 = rotl64_(k2, 33); k2 *= c1; h2 ^= k2;
        h2 = rotl64_(h2, 31); h2 += h1; h2 = h2 * 5 + 0x38495ab5;
    }
    const int8_t* tail = (const int8_t*)(key + nblocks * 16);
    uint64_t k1 = 0, k2 = 0;
    switch (length & 15) {  // SIGN-EXTENDED tail bytes (the "sign bug")
        case 15: k2 ^= ((uint64_t)(int64_t)tail[14]) << 48; [[fallthrough]];
        case 14: k2 ^= ((uint64_t)(int64_t)tail[13]) << 40; [[fallthrough]];
        case 13: k2 ^= ((uint64_t)(int64_t)tail[12]) << 32; [[fallthrough]];
        case 12: k2 ^= ((uint64_t)(int64_t)tail[11]) << 24; [[fallthrough]];
        case 11: k2 ^= ((uint64_t)(int64_t)tail[10]) << 16; [[fallthrough]];
        case 10: k2 ^= ((uint64_t)(int64_t)tail[9]) << 8; [[fallthrough]];
        case 9:  k2 ^= ((uint64_t)(int64_t)tail[8]);
                 k2 *= c2; k2 = rotl64_(k2, 33); k2 *= c1; h2 ^= k2; [[fallthrough]];
        case 8:  k1 ^= ((uint64_t)(int64_t)tail[7]) << 56; [[fallthrough]];
        case 7:  k1 ^= ((uint64_t)(int64_t)tail[6]) << 48; [[fallthrough]];
        case 6:  k1 ^= ((uint64_t)(int64_t)tail[5]) << 40; [[fallthrough]];
        case 5:  k1 ^= ((uint64_t)(int64_t)tail[4]) << 32; [[fallthrough]];
        case 4:  k1 ^= ((uint64_t)(int64_t)tail[3]) << 24; [[fallthrough]];
        case 3:  k1 ^= ((uint64_t)(int64_t)tail[2]) << 16; [[fallthrough]];
        case 2:  k1 ^= ((uint64_t)(int64_t)tail[1]) << 8; [[fallthrough]];
        case 1:  k1 ^= ((uint64_t)(int64_t)tail[0]);
                 k1 *= c1; k1 = rotl64_(k1, 31); k1 *= c2; h1 ^= k1; break;
        case 0: break;
    }
    h1 ^= (uint64_t)length; h2 ^= (uint64_t)length;
    h1 += h2; h2 += h1;
    h1 = fmix_(h1); h2 = fmix_(h2);
    h1 += h2; h2 += h1;
    out[0] = h1; out[1] = h2;
}

// Murmur3Partitioner token: normalize(hash[0]) — Murmur3Partitioner.java:256-295.
// Empty key -> MINIMUM token (Long.MIN_VALUE).
inline int64_t murmur3_token(const uint8_t* key, size_t len) {
    if (len == 0) return INT64_MIN;
    uint64_t h[2];
    murmur3_128_cassandra(key, len, 0, h);
    int64_t v = (int64_t)h[0];
    return v == INT64_MIN ? INT64_MAX : v;  // normalize (:291-295)
}

// DecoratedKey order: token, then unsigned key-byte compare
// (DecoratedKey.compareTo, DecoratedKey.java:79-92).
inline int compare_decorated_key(int64_t tok_a, const uint8_t* a, size_t alen,
                                 int64_t tok_b, const uint8_t* b, size_t blen) {
    if (tok_a != tok_b) return tok_a < tok_b ? -1 : 1;
    size_t n = alen < blen ? alen : blen;
    int c = memcmp(a, b, n);
    if (c) return c;
    return alen == blen ? 0 : (alen < blen ? -1 : 1);
}

// CRC32 (IEEE; Java java.util.zip.CRC32 == zlib crc32). Implemented locally to
// avoid a zlib link dependency in the kernelless oracle.
inline uint32_t crc32_update(uint32_t crc, const uint8_t* buf, size_t len) {
    static uint32_t table[256];
    static bool init = false;
    if (!init) {
        for (uint32_t i = 0; i < 256; i++) {
            uint32_t c = i;
            for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
            table[i] = c;
        }
        init = true;
    }
    crc = ~crc;
    for (size_t i = 0; i < len; i++) crc = table[(crc ^ buf[i]) & 0xFF] ^ (crc >> 8);
    return ~crc;
}
inline uint32_t crc32(const uint8_t* buf, size_t len) { return crc32_update(0, buf, len); }
// Java CRC32.update(int) on a single int (updateChecksumInt): big-endian 4 bytes.
inline uint32_t crc32_update_int(uint32_t crc, uint32_t v) {
    uint8_t b[4] = {(uint8_t)(v >> 24), (uint8_t)(v >> 16), (uint8_t)(v >> 8), (uint8_t)v};
    return crc32_update(crc, b, 4);
}

// splitmix64 — deterministic RNG for the synthetic generator (shared contract
// with the GPU generator; both sides derive identical rows from (seed, index)).
inline uint64_t splitmix64(uint64_t x) {
    x += 0x9E3779B97f4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

inline bytes read_file(const std::string& path) {
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) throw std::runtime_error("cannot open " + path);
    fseek(f, 0, SEEK_END);
    long n = ftell(f);
    fseek(f, 0, SEEK_SET);
    bytes b((size_t)n);
    if (n && fread(b.data(), 1, (size_t)n, f) != (size_t)n) { fclose(f); throw std::runtime_error("short file read " + path); }
    fclose(f);
    return b;
}
inline void write_file(const std::string& path, const bytes& b) {
    FILE* f = fopen(path.c_str(), "wb");
    if (!f) throw std::runtime_error("cannot create " + path);
    if (b.size() && fwrite(b.data(), 1, b.size(), f) != b.size()) { fclose(f); throw std::runtime_error("short write " + path); }
    fclose(f);
}


// Streaming MurmurHash3 x64_128 with integer seed (Guava Murmur3_128Hasher
// semantics: h1 = h2 = seed; finalize XORs total length; output h1||h2
// little-endian). The repair Validator's digest is
// concat(murmur3_128(1000), murmur3_128(2000)) over the same byte stream
// (db/Digest.java:53-59).
struct M3Stream {
    uint64_t h1, h2, len;
    uint8_t buf[16];
    uint32_t n;
    void init(int64_t seed) {
        h1 = (uint64_t)seed;
        h2 = (uint64_t)seed;
        len = 0;
        n = 0;
    }
    static uint64_t rotl(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }
    void block(const uint8_t* p) {
        const uint64_t c1 = 0x87c37b91114253d5ULL, c2 = 0x4cf5ad432745937fULL;
        uint64_t k1, k2;
        memcpy(&k1, p, 8);
        memcpy(&k2, p + 8, 8);
        k1 *= c1; k1 = rotl(k1, 31); k1 *= c2; h1 ^= k1;
        h1 = rotl(h1, 27); h1 += h2; h1 = h1 * 5 + 0x52dce729;
        k2 *= c2; k2 = rotl(k2, 33); k2 *= c1; h2 ^= k2;
        h2 = rotl(h2, 31); h2 += h1; h2 = h2 * 5 + 0x38495ab5;
    }
    void put(const uint8_t* p, uint64_t m) {
        len += m;
        if (n) {
            while (m && n < 16) { buf[n++] = *p++; m--; }
            if (n == 16) { block(buf); n = 0; }
        }
        while (m >= 16) { block(p); p += 16; m -= 16; }
        while (m) { buf[n++] = *p++; m--; }
    }
    void put_u8(uint8_t v) { put(&v, 1); }
    void put_i32be(int32_t v) {  // Digest.updateWithInt
        uint8_t b[4];
        for (int i = 0; i < 4; i++) b[i] = (uint8_t)((uint32_t)v >> (8 * (3 - i)));
        put(b, 4);
    }
    void put_i64be(int64_t v) {  // Digest.updateWithLong
        uint8_t b[8];
        for (int i = 0; i < 8; i++) b[i] = (uint8_t)((uint64_t)v >> (8 * (7 - i)));
        put(b, 8);
    }
    void put_bool(bool v) { put_u8(v ? 0 : 1); }  // updateWithBoolean INVERTS
    static uint64_t fmix(uint64_t k) {
        k ^= k >> 33;
        k *= 0xff51afd7ed558ccdULL;
        k ^= k >> 33;
        k *= 0xc4ceb9fe1a85ec53ULL;
        k ^= k >> 33;
        return k;
    }
    void final16(uint8_t out[16]) {
        const uint64_t c1 = 0x87c37b91114253d5ULL, c2 = 0x4cf5ad432745937fULL;
        uint64_t k1 = 0, k2 = 0;
        if (n > 8) {
            for (uint32_t i = n; i > 8;) { i--; k2 = (k2 << 8) | buf[i]; }
            k2 *= c2; k2 = rotl(k2, 33); k2 *= c1; h2 ^= k2;
        }
        if (n > 0) {
            uint32_t m = n > 8 ? 8 : n;
            for (uint32_t i = m; i > 0;) { i--; k1 = (k1 << 8) | buf[i]; }
            k1 *= c1; k1 = rotl(k1, 31); k1 *= c2; h1 ^= k1;
        }
        h1 ^= len; h2 ^= len;
        h1 += h2; h2 += h1;
        h1 = fmix(h1); h2 = fmix(h2);
        h1 += h2; h2 += h1;
        memcpy(out, &h1, 8);       // little-endian out (Guava asBytes)
        memcpy(out + 8, &h2, 8);
    }
};

}  // namespace oracle
