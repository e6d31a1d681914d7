// Scalar restatement of snappy 1.1.8's block compressor (snappy.cc
// CompressFragment + the framing in RawCompress), the codec the reference
// uses through snappy-java 1.1.10.4 (SnappyCompressor.java:82-86; see the
// 1.1.8-vs-1.1.10 pin caveat in BASELINE.md). Shared by the CPU oracle pin
// test, the wave simulator and the HIP kernel: this file is the SPEC the
// wave implementation must match byte-for-byte, and it is itself pinned
// against the system libsnappy by tests/test_snappy_model (CPU).
//
// Layout of one compressed block (format_description.txt):
//   varint32 uncompressed_length, then tagged elements:
//     tag&3==0: literal, len = (tag>>2)+1 (<=60) or 1/2/3/4 extra LE bytes
//     tag&3==1: copy1 — len 4..11 = ((tag>>2)&7)+4, offset 11 bit = ((tag>>5)<<8)|next
//     tag&3==2: copy2 — len = (tag>>2)+1, offset 16-bit LE
#pragma once
#include <cstdint>
#include <cstring>

#if defined(__HIPCC__)
#define SNP_HD __host__ __device__
#else
#define SNP_HD
#endif

namespace gpuc {

constexpr uint32_t SNP_BLOCK_LOG = 16;              // kBlockLog (64 KiB fragments)
constexpr uint32_t SNP_MAX_TABLE = 1u << 14;        // kMaxHashTableSize entries
constexpr uint32_t SNP_INPUT_MARGIN = 15;           // kInputMarginBytes

SNP_HD inline uint32_t snp_load32(const uint8_t* p) { uint32_t v; memcpy(&v, p, 4); return v; }
SNP_HD inline uint64_t snp_load64(const uint8_t* p) { uint64_t v; memcpy(&v, p, 8); return v; }

SNP_HD inline uint32_t snp_hash(uint32_t bytes, int shift) {
    return (bytes * 0x1e35a7bdu) >> shift;
}

// smallest power of two >= max(16, min(input, kMaxHashTableSize)) — snappy.cc
// Alloc/WorkingMemory::GetHashTable semantics
SNP_HD inline uint32_t snp_table_size(uint32_t input_size) {
    uint32_t htsize = 256;
    while (htsize < SNP_MAX_TABLE && htsize < input_size) htsize <<= 1;
    return htsize;
}

inline uint8_t* snp_emit_literal(uint8_t* op, const uint8_t* literal, int len) {
    int n = len - 1;
    if (n < 60) {
        *op++ = (uint8_t)(n << 2);
    } else {
        uint8_t* base = op;
        op++;
        int count = 0;
        while (n > 0) { *op++ = (uint8_t)(n & 0xff); n >>= 8; count++; }
        *base = (uint8_t)((59 + count) << 2);
    }
    memcpy(op, literal, (size_t)len);
    return op + len;
}

inline uint8_t* snp_emit_copy_upto64(uint8_t* op, size_t offset, int len) {
    if (len < 12 && offset < 2048) {
        *op++ = (uint8_t)(1 | ((len - 4) << 2) | ((offset >> 8) << 5));
        *op++ = (uint8_t)(offset & 0xff);
    } else {
        *op++ = (uint8_t)(2 | ((len - 1) << 2));
        *op++ = (uint8_t)(offset & 0xff);
        *op++ = (uint8_t)(offset >> 8);
    }
    return op;
}

inline uint8_t* snp_emit_copy(uint8_t* op, size_t offset, int len) {
    while (len >= 68) {
        op = snp_emit_copy_upto64(op, offset, 64);
        len -= 64;
    }
    if (len > 64) {
        op = snp_emit_copy_upto64(op, offset, 60);
        len -= 60;
    }
    return snp_emit_copy_upto64(op, offset, len);
}

// FindMatchLength(s1, s2, s2_limit): bytes equal starting at s1/s2, s2 bounded
inline int snp_match_length(const uint8_t* s1, const uint8_t* s2, const uint8_t* s2_limit) {
    int matched = 0;
    while (s2 + 8 <= s2_limit && snp_load64(s2) == snp_load64(s1 + matched)) {
        s2 += 8;
        matched += 8;
    }
    while (s2 < s2_limit && s1[matched] == *s2) {
        s2++;
        matched++;
    }
    return matched;
}

// CompressFragment (snappy.cc 1.1.8), one fragment (input < 64 KiB here:
// Cassandra chunks are 16 KiB). table: uint16[snp_table_size], caller-zeroed.
inline int snp_compress_fragment(const uint8_t* input, uint32_t input_size, uint8_t* op0,
                                 uint16_t* table, uint32_t table_size) {
    uint8_t* op = op0;
    const int shift = 32 - __builtin_ctz(table_size);  // table_size is a power of 2
    const uint8_t* ip = input;
    const uint8_t* ip_end = input + input_size;
    const uint8_t* base_ip = input;
    const uint8_t* next_emit = input;

    if (input_size >= SNP_INPUT_MARGIN) {
        const uint8_t* ip_limit = input + input_size - SNP_INPUT_MARGIN;
        for (uint32_t next_hash = snp_hash(snp_load32(++ip), shift);;) {
            uint32_t skip = 32;
            const uint8_t* next_ip = ip;
            const uint8_t* candidate;
            do {
                ip = next_ip;
                uint32_t hash = next_hash;
                uint32_t bytes_between_hash_lookups = skip >> 5;
                skip += bytes_between_hash_lookups;
                next_ip = ip + bytes_between_hash_lookups;
                if (next_ip > ip_limit) goto emit_remainder;
                next_hash = snp_hash(snp_load32(next_ip), shift);
                candidate = base_ip + table[hash];
                table[hash] = (uint16_t)(ip - base_ip);
            } while (snp_load32(ip) != snp_load32(candidate));

            op = snp_emit_literal(op, next_emit, (int)(ip - next_emit));

            uint64_t input_bytes = 0;
            uint32_t candidate_bytes = 0;
            do {
                const uint8_t* base = ip;
                int matched = 4 + snp_match_length(candidate + 4, ip + 4, ip_end);
                ip += matched;
                size_t offset = (size_t)(base - candidate);
                op = snp_emit_copy(op, offset, matched);
                next_emit = ip;
                if (ip >= ip_limit) goto emit_remainder;
                // GetEightBytesAt(ip - 1): insert ip-1 and probe ip
                input_bytes = snp_load64(ip - 1);
                uint32_t prev_hash = snp_hash((uint32_t)input_bytes, shift);
                table[prev_hash] = (uint16_t)(ip - base_ip - 1);
                uint32_t cur_hash = snp_hash((uint32_t)(input_bytes >> 8), shift);
                candidate = base_ip + table[cur_hash];
                candidate_bytes = snp_load32(candidate);
                table[cur_hash] = (uint16_t)(ip - base_ip);
            } while ((uint32_t)(input_bytes >> 8) == candidate_bytes);
            next_hash = snp_hash((uint32_t)(input_bytes >> 16), shift);
            ip++;
        }
    }
emit_remainder:
    if (next_emit < ip_end)
        op = snp_emit_literal(op, next_emit, (int)(ip_end - next_emit));
    return (int)(op - op0);
}

// RawCompress framing for one Cassandra chunk (< kBlockSize, one fragment):
// varint32 length prefix + fragment
inline int snp_compress(const uint8_t* input, uint32_t n, uint8_t* out,
                        uint16_t* table, uint32_t table_size) {
    uint8_t* op = out;
    uint32_t v = n;
    while (v >= 0x80) { *op++ = (uint8_t)(v | 0x80); v >>= 7; }
    *op++ = (uint8_t)v;
    // snappy zeroes the table per fragment (WorkingMemory memset)
    memset(table, 0, table_size * sizeof(uint16_t));
    int frag = snp_compress_fragment(input, n, op, table, table_size);
    return (int)(op - out) + frag;
}

}  // namespace gpuc
