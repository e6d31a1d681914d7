// PRODUCT — host+device codec primitives for the MI355X compaction library.
// Independent restatement of the reference encodings (citations inline);
// deliberately NOT shared with oracle/ (the oracle is the checker).
#pragma once
#include <stdint.h>
#include <string.h>

#if defined(__HIPCC__)
#define GPUC_HD __host__ __device__
#else
#define GPUC_HD
#endif

namespace gpuc {

// ---- constants (db/LivenessInfo.java, db/rows/Cell.java, db/rows/EncodingStats.java) ----
constexpr int64_t NO_TIMESTAMP = INT64_MIN;
constexpr int32_t NO_TTL = 0;
constexpr int64_t NO_DELETION_TIME = INT64_MAX;
constexpr uint32_t LDT_NONE_U32 = 0xFFFFFFFFu;
constexpr int64_t TIMESTAMP_EPOCH = 1442880000000000LL;
constexpr int64_t DELETION_TIME_EPOCH = 1442880000LL;
constexpr uint32_t CHUNK_LEN = 16384;  // CompressionParams.DEFAULT_CHUNK_LENGTH

GPUC_HD inline int64_t ldt_long(uint32_t u) { return u == LDT_NONE_U32 ? NO_DELETION_TIME : (int64_t)u; }
GPUC_HD inline uint32_t ldt_u32(int64_t l) { return l == NO_DELETION_TIME ? LDT_NONE_U32 : (uint32_t)l; }

// ---- unsigned vint (utils/vint/VIntCoding.java) ----
GPUC_HD inline int uvint_size(uint64_t v) {
#if defined(__HIP_DEVICE_COMPILE__)
    int magnitude = __clzll((long long)(v | 1));
#else
    int magnitude = __builtin_clzll(v | 1);
#endif
    return (639 - magnitude * 9) >> 6;
}
// emit into p, return bytes written
GPUC_HD inline int uvint_put(uint8_t* p, uint64_t v) {
    int size = uvint_size(v);
    if (size == 1) { p[0] = (uint8_t)v; return 1; }
    if (size == 9) {
        p[0] = 0xFF;
        for (int i = 0; i < 8; i++) p[1 + i] = (uint8_t)(v >> (8 * (7 - i)));
        return 9;
    }
    int extra = size - 1;
    uint64_t reg = v << ((8 - size) * 8);
    p[0] = (uint8_t)((reg >> 56) | (uint8_t)~(0xFFu >> extra));
    for (int i = 1; i < size; i++) p[i] = (uint8_t)(reg >> (8 * (7 - i)));
    return size;
}
// read; advances *pos
GPUC_HD inline uint64_t uvint_get(const uint8_t* p, uint64_t* pos) {
    int8_t first = (int8_t)p[(*pos)++];
    if (first >= 0) return (uint64_t)first;
    uint32_t inv = ~(uint32_t)(int32_t)first;
#if defined(__HIP_DEVICE_COMPILE__)
    int extra = inv == 0 ? 8 : __clz((int)inv) - 24;
#else
    int extra = inv == 0 ? 8 : __builtin_clz(inv) - 24;
#endif
    uint64_t v = (uint8_t)first & (0xFFu >> extra);
    for (int i = 0; i < extra; i++) v = (v << 8) | p[(*pos)++];
    return v;
}
GPUC_HD inline uint64_t zigzag(int64_t n) { return ((uint64_t)n << 1) ^ (uint64_t)(n >> 63); }
GPUC_HD inline int64_t unzigzag(uint64_t n) { return (int64_t)(n >> 1) ^ -(int64_t)(n & 1); }

// sign-extended u32 diff per writeUnsignedVInt32 (VIntCoding.java:329-332)
GPUC_HD inline uint64_t sext32(int64_t diff) { return (uint64_t)(int64_t)(int32_t)diff; }

// ---- MurmurHash.hash3_x64_128 with Cassandra's sign-extended tail
//      (utils/MurmurHash.java:178-253) ----
GPUC_HD inline uint64_t mm_rotl(uint64_t v, int n) { return (v << n) | (v >> (64 - n)); }
GPUC_HD inline uint64_t mm_fmix(uint64_t k) {
    k ^= k >> 33; k *= 0xff51afd7ed558ccdULL;
    k ^= k >> 33; k *= 0xc4ceb9fe1a85ec53ULL;
    k ^= k >> 33; return k;
}
GPUC_HD inline void murmur3_128(const uint8_t* key, uint32_t length, uint64_t seed, uint64_t out[2]) {
    const uint32_t nblocks = length >> 4;
    uint64_t h1 = seed, h2 = seed;
    const uint64_t c1 = 0x87c37b91114253d5ULL, c2 = 0x4cf5ad432745937fULL;
    for (uint32_t i = 0; i < nblocks; i++) {
        uint64_t k1, k2;
        memcpy(&k1, key + i * 16, 8);
        memcpy(&k2, key + i * 16 + 8, 8);
        k1 *= c1; k1 = mm_rotl(k1, 31); k1 *= c2; h1 ^= k1;
        h1 = mm_rotl(h1, 27); h1 += h2; h1 = h1 * 5 + 0x52dce729;
        k2 *= c2; k2 = mm_rotl(k2, 33); k2 *= c1; h2 ^= k2;
        h2 = mm_rotl(h2, 31); h2 += h1; h2 = h2 * 5 + 0x38495ab5;
    }
    const int8_t* tail = (const int8_t*)(key + nblocks * 16);
    uint64_t k1 = 0, k2 = 0;
    switch (length & 15) {  // sign-extended tail bytes (the historical sign bug)
        case 15: k2 ^= ((uint64_t)(int64_t)tail[14]) << 48; [[fallthrough]];
        case 14: k2 ^= ((uint64_t)(int64_t)tail[13]) << 40; [[fallthrough]];
        case 13: k2 ^= ((uint64_t)(int64_t)tail[12]) << 32; [[fallthrough]];
        case 12: k2 ^= ((uint64_t)(int64_t)tail[11]) << 24; [[fallthrough]];
        case 11: k2 ^= ((uint64_t)(int64_t)tail[10]) << 16; [[fallthrough]];
        case 10: k2 ^= ((uint64_t)(int64_t)tail[9]) << 8; [[fallthrough]];
        case 9:  k2 ^= ((uint64_t)(int64_t)tail[8]);
                 k2 *= c2; k2 = mm_rotl(k2, 33); k2 *= c1; h2 ^= k2; [[fallthrough]];
        case 8:  k1 ^= ((uint64_t)(int64_t)tail[7]) << 56; [[fallthrough]];
        case 7:  k1 ^= ((uint64_t)(int64_t)tail[6]) << 48; [[fallthrough]];
        case 6:  k1 ^= ((uint64_t)(int64_t)tail[5]) << 40; [[fallthrough]];
        case 5:  k1 ^= ((uint64_t)(int64_t)tail[4]) << 32; [[fallthrough]];
        case 4:  k1 ^= ((uint64_t)(int64_t)tail[3]) << 24; [[fallthrough]];
        case 3:  k1 ^= ((uint64_t)(int64_t)tail[2]) << 16; [[fallthrough]];
        case 2:  k1 ^= ((uint64_t)(int64_t)tail[1]) << 8; [[fallthrough]];
        case 1:  k1 ^= ((uint64_t)(int64_t)tail[0]);
                 k1 *= c1; k1 = mm_rotl(k1, 31); k1 *= c2; h1 ^= k1; break;
        default: break;
    }
    h1 ^= length; h2 ^= length;
    h1 += h2; h2 += h1;
    h1 = mm_fmix(h1); h2 = mm_fmix(h2);
    h1 += h2; h2 += h1;
    out[0] = h1; out[1] = h2;
}
// Murmur3Partitioner token (Murmur3Partitioner.java:256-295)
GPUC_HD inline int64_t murmur3_token(const uint8_t* key, uint32_t len) {
    if (len == 0) return INT64_MIN;
    uint64_t h[2];
    murmur3_128(key, len, 0, h);
    int64_t v = (int64_t)h[0];
    return v == INT64_MIN ? INT64_MAX : v;
}

// ---- CRC32 (IEEE, == java.util.zip.CRC32) ----
// bitwise (slow, used for small host pieces and as device fallback)
GPUC_HD inline uint32_t crc32_update_bitwise(uint32_t crc, const uint8_t* buf, size_t len) {
    crc = ~crc;
    for (size_t i = 0; i < len; i++) {
        crc ^= buf[i];
        for (int k = 0; k < 8; k++) crc = (crc & 1) ? 0xEDB88320u ^ (crc >> 1) : crc >> 1;
    }
    return ~crc;
}
// table-driven update against caller-provided 256-entry table
GPUC_HD inline uint32_t crc32_update_t(uint32_t crc, const uint8_t* buf, size_t len,
                                       const uint32_t* table) {
    crc = ~crc;
    for (size_t i = 0; i < len; i++) crc = table[(crc ^ buf[i]) & 0xFF] ^ (crc >> 8);
    return ~crc;
}
// slicing-by-8: t[k][b] = table-walk of byte b followed by k zero bytes
inline void crc32_make_table8(uint32_t* t /* [8*256] */) {
    for (uint32_t i = 0; i < 256; i++) {
        uint32_t c = i;
        for (int j = 0; j < 8; j++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
        t[i] = c;
    }
    for (int k = 1; k < 8; k++)
        for (uint32_t i = 0; i < 256; i++)
            t[k * 256 + i] = t[(t[(k - 1) * 256 + i] & 0xFF)] ^ (t[(k - 1) * 256 + i] >> 8);
}
inline void crc32_make_table(uint32_t* table) {
    for (uint32_t i = 0; i < 256; i++) {
        uint32_t c = i;
        for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
        table[i] = c;
    }
}
// crc32_combine (zlib algorithm): crc(A||B) from crc(A), crc(B), len(B)
inline uint32_t gf2_times(const uint32_t* mat, uint32_t vec) {
    uint32_t sum = 0;
    int i = 0;
    while (vec) {
        if (vec & 1) sum ^= mat[i];
        vec >>= 1;
        i++;
    }
    return sum;
}
inline void gf2_square(uint32_t* sq, const uint32_t* mat) {
    for (int n = 0; n < 32; n++) sq[n] = gf2_times(mat, mat[n]);
}
struct Crc32Combiner {
    uint32_t pow[64][32];  // pow[k] = operator for 2^k zero BYTES... (bits handled below)
    Crc32Combiner() {
        uint32_t odd[32], even[32];
        odd[0] = 0xEDB88320u;  // polynomial
        uint32_t row = 1;
        for (int n = 1; n < 32; n++) { odd[n] = row; row <<= 1; }
        gf2_square(even, odd);   // 2 zero bits
        gf2_square(odd, even);   // 4 zero bits
        gf2_square(even, odd);   // 8 zero bits = 1 zero byte
        memcpy(pow[0], even, sizeof(even));
        for (int k = 1; k < 64; k++) gf2_square(pow[k], pow[k - 1]);
    }
    uint32_t combine(uint32_t crc1, uint32_t crc2, uint64_t len2) const {
        if (len2 == 0) return crc1;
        uint32_t c = crc1;
        uint64_t n = len2;
        int k = 0;
        while (n) {
            if (n & 1) c = gf2_times(pow[k], c);
            n >>= 1;
            k++;
        }
        return c ^ crc2;
    }
};

// ---- splitmix64 / Feistel (shared generator contract, oracle/src/gen.h) ----
GPUC_HD inline uint64_t splitmix64(uint64_t x) {
    x += 0x9E3779B97f4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}
GPUC_HD inline uint64_t feistel_perm(uint64_t seed, uint64_t universe, uint64_t x) {
    int hb = 1;
    while ((1ULL << (2 * hb)) < universe) hb++;
    uint64_t mask = (1ULL << hb) - 1;
    do {
        uint64_t l = (x >> hb) & mask, r = x & mask;
        for (int round = 0; round < 4; round++) {
            uint64_t f = splitmix64(seed ^ r ^ ((uint64_t)(round + 1) << 56)) & mask;
            uint64_t nl = r;
            r = l ^ f;
            l = nl;
        }
        x = (l << hb) | r;
    } while (x >= universe);
    return x;
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
    GPUC_HD void init(int64_t seed) {
        h1 = (uint64_t)seed;
        h2 = (uint64_t)seed;
        len = 0;
        n = 0;
    }
    GPUC_HD static uint64_t rotl(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }
    GPUC_HD void block(const uint8_t* p) {
        const uint64_t c1 = 0x87c37b91114253d5ULL, c2 = 0x4cf5ad432745937fULL;
        uint64_t k1, k2;
        memcpy(&k1, p, 8);
        memcpy(&k2, p + 8, 8);
        k1 *= c1; k1 = rotl(k1, 31); k1 *= c2; h1 ^= k1;
        h1 = rotl(h1, 27); h1 += h2; h1 = h1 * 5 + 0x52dce729;
        k2 *= c2; k2 = rotl(k2, 33); k2 *= c1; h2 ^= k2;
        h2 = rotl(h2, 31); h2 += h1; h2 = h2 * 5 + 0x38495ab5;
    }
    GPUC_HD void put(const uint8_t* p, uint64_t m) {
        len += m;
        if (n) {
            while (m && n < 16) { buf[n++] = *p++; m--; }
            if (n == 16) { block(buf); n = 0; }
        }
        while (m >= 16) { block(p); p += 16; m -= 16; }
        while (m) { buf[n++] = *p++; m--; }
    }
    GPUC_HD void put_u8(uint8_t v) { put(&v, 1); }
    GPUC_HD void put_i32be(int32_t v) {  // Digest.updateWithInt
        uint8_t b[4];
        for (int i = 0; i < 4; i++) b[i] = (uint8_t)((uint32_t)v >> (8 * (3 - i)));
        put(b, 4);
    }
    GPUC_HD void put_i64be(int64_t v) {  // Digest.updateWithLong
        uint8_t b[8];
        for (int i = 0; i < 8; i++) b[i] = (uint8_t)((uint64_t)v >> (8 * (7 - i)));
        put(b, 8);
    }
    GPUC_HD void put_bool(bool v) { put_u8(v ? 0 : 1); }  // updateWithBoolean INVERTS
    GPUC_HD static uint64_t fmix(uint64_t k) {
        k ^= k >> 33;
        k *= 0xff51afd7ed558ccdULL;
        k ^= k >> 33;
        k *= 0xc4ceb9fe1a85ec53ULL;
        k ^= k >> 33;
        return k;
    }
    GPUC_HD void final16(uint8_t out[16]) {
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

}  // namespace gpuc
