// PRODUCT — bit-exact restatement of LZ4_compress_default (liblz4 1.9.3,
// byU16 strategy: input < 64 KB, acceleration 1) as a single-thread routine
// compilable as both host and HIP device code. The GPU chunk-compress kernel
// runs this per 16 KiB chunk (one per wave; lane 0 drives control flow, the
// wave assists bulk compares/copies in the .hip wrapper).
//
// The reference's compression bytes come from lz4-java 1.8.0, which bundles
// liblz4 1.9.3 == this image's system liblz4 (BASELINE.md "parity anchors").
// tests/test_lz4_model.py fuzzes this model against the system library.
//
// Cassandra chunk framing context: LZ4Compressor.compress writes a 4-byte
// little-endian uncompressed length then this block (LZ4Compressor.java:118-134).
#pragma once
#include <stdint.h>
#include <string.h>

#ifdef __HIP_DEVICE_COMPILE__
#define LZ4M_HD __host__ __device__
#elif defined(__HIPCC__)
#define LZ4M_HD __host__ __device__
#else
#define LZ4M_HD
#endif

// constants from lz4.c 1.9.3
#define LZ4M_MINMATCH 4
#define LZ4M_MFLIMIT 12
#define LZ4M_LASTLITERALS 5
#define LZ4M_SKIPTRIGGER 6
#define LZ4M_HASHLOG 12                 // LZ4_MEMORY_USAGE(14) - 2
#define LZ4M_HASHTABLESIZE_U16 (1 << (LZ4M_HASHLOG + 1))  // 8192 entries (byU16)
#define LZ4M_ML_BITS 4
#define LZ4M_ML_MASK ((1U << LZ4M_ML_BITS) - 1)
#define LZ4M_RUN_BITS (8 - LZ4M_ML_BITS)
#define LZ4M_RUN_MASK ((1U << LZ4M_RUN_BITS) - 1)

LZ4M_HD static inline uint32_t lz4m_read32(const uint8_t* p) {
    uint32_t v;
    memcpy(&v, p, 4);
    return v;  // little-endian host/device
}
LZ4M_HD static inline uint16_t lz4m_read16(const uint8_t* p) {
    uint16_t v;
    memcpy(&v, p, 2);
    return v;
}
LZ4M_HD static inline uint64_t lz4m_read64(const uint8_t* p) {
    uint64_t v;
    memcpy(&v, p, 8);
    return v;
}
// LZ4_hash4 for byU16: (seq * 2654435761U) >> ((MINMATCH*8) - (HASHLOG+1))
LZ4M_HD static inline uint32_t lz4m_hash(uint32_t sequence) {
    return (sequence * 2654435761U) >> ((LZ4M_MINMATCH * 8) - (LZ4M_HASHLOG + 1));
}
// LZ4_NbCommonBytes on little-endian 64-bit: count of matching leading bytes
LZ4M_HD static inline unsigned lz4m_nb_common_bytes(uint64_t diff) {
#if defined(__HIP_DEVICE_COMPILE__)
    return (unsigned)(__ffsll((long long)diff) - 1) >> 3;
#else
    return (unsigned)__builtin_ctzll(diff) >> 3;
#endif
}
// LZ4_count: number of common bytes from pIn/pMatch up to pInLimit
LZ4M_HD static inline unsigned lz4m_count(const uint8_t* pIn, const uint8_t* pMatch,
                                          const uint8_t* pInLimit) {
    const uint8_t* const pStart = pIn;
    while (pIn < pInLimit - (8 - 1)) {
        uint64_t diff = lz4m_read64(pMatch) ^ lz4m_read64(pIn);
        if (!diff) { pIn += 8; pMatch += 8; continue; }
        return (unsigned)(pIn + lz4m_nb_common_bytes(diff) - pStart);
    }
    if ((pIn < pInLimit - 3) && (lz4m_read32(pMatch) == lz4m_read32(pIn))) { pIn += 4; pMatch += 4; }
    if ((pIn < pInLimit - 1) && (lz4m_read16(pMatch) == lz4m_read16(pIn))) { pIn += 2; pMatch += 2; }
    if ((pIn < pInLimit) && (*pMatch == *pIn)) pIn++;
    return (unsigned)(pIn - pStart);
}

// LZ4_compress_generic(ctx, src, dst, srcSize, NULL, 0, notLimited, byU16,
// noDict, noDictIssue, acceleration=1) — lz4.c:807-1028, restated for the
// byU16/noDict/notLimited path only. `table` is caller-provided u16[8192],
// zero-initialised (LZ4_prepareTable memsets for a fresh state).
// Returns compressed size. srcSize must be >= 1 and < 65536 (byU16 regime)
// and dst must have LZ4_compressBound(srcSize) capacity.
LZ4M_HD static inline int lz4m_compress(const uint8_t* src, int srcSize, uint8_t* dst,
                                        uint16_t* table) {
    const uint8_t* ip = src;
    const uint8_t* anchor = src;
    const uint8_t* const iend = src + srcSize;
    const uint8_t* const mflimitPlusOne = iend - LZ4M_MFLIMIT + 1;
    const uint8_t* const matchlimit = iend - LZ4M_LASTLITERALS;
    uint8_t* op = dst;
    uint32_t forwardH;

    if (srcSize < LZ4M_MFLIMIT + 1) goto _last_literals;  // lz4.c: skip search for tiny inputs

    // First byte
    table[lz4m_hash(lz4m_read32(ip))] = (uint16_t)(ip - src);
    ip++;
    forwardH = lz4m_hash(lz4m_read32(ip));

    for (;;) {
        const uint8_t* match;
        uint8_t* token;
        // --- find a match (acceleration=1) ---
        {
            const uint8_t* forwardIp = ip;
            int step = 1;
            int searchMatchNb = 1 << LZ4M_SKIPTRIGGER;  // acceleration << skipTrigger
            do {
                uint32_t const h = forwardH;
                ip = forwardIp;
                forwardIp += step;
                step = (searchMatchNb++ >> LZ4M_SKIPTRIGGER);
                if (forwardIp > mflimitPlusOne) goto _last_literals;
                match = src + table[h];
                forwardH = lz4m_hash(lz4m_read32(forwardIp));
                table[h] = (uint16_t)(ip - src);
            } while (lz4m_read32(match) != lz4m_read32(ip));
            // (byU16: no maxDistance check — whole input within 64 KB window)
        }

        // --- catch up ---
        while (((ip > anchor) & (match > src)) && (ip[-1] == match[-1])) { ip--; match--; }

        // --- encode literals ---
        {
            unsigned litLength = (unsigned)(ip - anchor);
            token = op++;
            if (litLength >= LZ4M_RUN_MASK) {
                int len = (int)(litLength - LZ4M_RUN_MASK);
                *token = (uint8_t)(LZ4M_RUN_MASK << LZ4M_ML_BITS);
                for (; len >= 255; len -= 255) *op++ = 255;
                *op++ = (uint8_t)len;
            } else {
                *token = (uint8_t)(litLength << LZ4M_ML_BITS);
            }
            memcpy(op, anchor, litLength);  // wildCopy8 equivalent for correctness
            op += litLength;
        }

    _next_match:
        // --- encode offset ---
        op[0] = (uint8_t)(ip - match);
        op[1] = (uint8_t)((ip - match) >> 8);
        op += 2;

        // --- encode match length ---
        {
            unsigned matchCode = lz4m_count(ip + LZ4M_MINMATCH, match + LZ4M_MINMATCH, matchlimit);
            ip += (size_t)matchCode + LZ4M_MINMATCH;
            if (matchCode >= LZ4M_ML_MASK) {
                // lz4.c writes 0xFFFFFFFF ahead then advances by matchCode/255
                *token += LZ4M_ML_MASK;
                matchCode -= LZ4M_ML_MASK;
                memset(op, 255, 4);
                while (matchCode >= 4 * 255) {
                    op += 4;
                    memset(op, 255, 4);
                    matchCode -= 4 * 255;
                }
                op += matchCode / 255;
                *op++ = (uint8_t)(matchCode % 255);
            } else {
                *token += (uint8_t)matchCode;
            }
        }

        anchor = ip;

        // --- test end of chunk ---
        if (ip >= mflimitPlusOne) break;

        // --- fill table ---
        table[lz4m_hash(lz4m_read32(ip - 2))] = (uint16_t)(ip - 2 - src);

        // --- test next position ---
        {
            uint32_t const h = lz4m_hash(lz4m_read32(ip));
            match = src + table[h];
            table[h] = (uint16_t)(ip - src);
            if (lz4m_read32(match) == lz4m_read32(ip)) { token = op++; *token = 0; goto _next_match; }
        }

        // --- prepare next loop ---
        forwardH = lz4m_hash(lz4m_read32(++ip));
    }

_last_literals:
    {
        size_t lastRun = (size_t)(iend - anchor);
        if (lastRun >= LZ4M_RUN_MASK) {
            size_t accumulator = lastRun - LZ4M_RUN_MASK;
            *op++ = (uint8_t)(LZ4M_RUN_MASK << LZ4M_ML_BITS);
            for (; accumulator >= 255; accumulator -= 255) *op++ = 255;
            *op++ = (uint8_t)accumulator;
        } else {
            *op++ = (uint8_t)(lastRun << LZ4M_ML_BITS);
        }
        memcpy(op, anchor, lastRun);
        op += lastRun;
    }
    return (int)(op - dst);
}
