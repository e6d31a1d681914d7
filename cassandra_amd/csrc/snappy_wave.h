// PRODUCT (round-2 integration pending) — wave-cooperative snappy kernels for
// CDNA4 (gfx950): bit-exact snappy 1.1.8 block compressor (the codec the
// reference uses via snappy-java; SnappyCompressor.java:82-86) and the block
// decompressor. The compressor runs the scalar probe chain (snappy.cc
// CompressFragment, restated in snappy_model.h and CPU-pinned against the
// system libsnappy) as 64-probe speculative windows with the same LDS
// marker / predecessor / restore+commit machinery as the LZ4 kernel
// (lz4_wave.h); probe positions come from the skip-heuristic offset table
// (skip=32; inc=skip>>5) passed as a kernel argument.
// Control flow mirrors tests/native/snappy_sim.h, which is proven byte-equal
// to the model on CPU.
#pragma once
#include <hip/hip_runtime.h>
#include "snappy_model.h"

namespace gpuc {

#ifndef WAVE
#define WAVE 64
#endif

__device__ inline int snp_wave_compress(const uint8_t* __restrict__ s, uint32_t n,
                                        uint8_t* __restrict__ dst,
                                        uint16_t* __restrict__ s_table, uint32_t table_size,
                                        const uint32_t* __restrict__ OFF, uint32_t off_n,
                                        int lane) {
    uint32_t op = 0;
    {   // varint length prefix (uniform; lane 0 stores)
        uint32_t v = n, i = 0;
        uint8_t tmp[5];
        while (v >= 0x80) { tmp[i++] = (uint8_t)(v | 0x80); v >>= 7; }
        tmp[i++] = (uint8_t)v;
        if (lane == 0)
            for (uint32_t j = 0; j < i; j++) dst[op + j] = tmp[j];
        op += i;
    }
    for (uint32_t i = lane; i < table_size; i += WAVE) s_table[i] = 0;
    // NOTE: caller must __syncthreads()-equivalently order this zeroing; one
    // wave per chunk => the wave's own LDS ops are in order, nothing needed.
    const int shift = 32 - __ffs((int)table_size) + 1;  // 32 - log2(table_size)
    const int ip_limit = (int)n - (int)SNP_INPUT_MARGIN;
    int ip = 1;
    int next_emit = 0;

    auto rd32g = [&](int p) { uint32_t v; memcpy(&v, s + p, 4); return v; };

    auto emit_literal = [&](int from, int len) {
        // tag (lane 0) + wave dword copy of the literal bytes
        int nm1 = len - 1;
        if (nm1 < 60) {
            if (lane == 0) dst[op] = (uint8_t)(nm1 << 2);
            op += 1;
        } else {
            uint8_t tmp[5];
            uint32_t cnt = 0, v2 = (uint32_t)nm1;
            while (v2 > 0) { tmp[cnt++] = (uint8_t)(v2 & 0xff); v2 >>= 8; }
            if (lane == 0) {
                dst[op] = (uint8_t)((59 + cnt) << 2);
                for (uint32_t j = 0; j < cnt; j++) dst[op + 1 + j] = tmp[j];
            }
            op += 1 + cnt;
        }
        for (int i = 4 * lane; i < len; i += 4 * WAVE) {
            int nb = len - i;
            if (nb >= 4) {
                uint32_t v;
                memcpy(&v, s + from + i, 4);
                memcpy(dst + op + i, &v, 4);
            } else {
                for (int j = 0; j < nb; j++) dst[op + i + j] = s[from + i + j];
            }
        }
        op += len;
    };
    auto emit_copy_upto64 = [&](uint32_t offset, int len) {
        if (len < 12 && offset < 2048) {
            if (lane == 0) {
                dst[op] = (uint8_t)(1 | ((len - 4) << 2) | ((offset >> 8) << 5));
                dst[op + 1] = (uint8_t)(offset & 0xff);
            }
            op += 2;
        } else {
            if (lane == 0) {
                dst[op] = (uint8_t)(2 | ((len - 1) << 2));
                dst[op + 1] = (uint8_t)(offset & 0xff);
                dst[op + 2] = (uint8_t)(offset >> 8);
            }
            op += 3;
        }
    };
    auto emit_copy = [&](uint32_t offset, int len) {
        while (len >= 68) { emit_copy_upto64(offset, 64); len -= 64; }
        if (len > 64) { emit_copy_upto64(offset, 60); len -= 60; }
        emit_copy_upto64(offset, len);
    };

    if ((int)n >= (int)SNP_INPUT_MARGIN) {
        while (true) {
            // ---- probe windows from run start `ip` ----
            int match = -1;
            {
                uint32_t k0 = 0;
                const int S0 = ip;
                bool found = false, aborted = false;
                while (true) {
                    uint32_t m_idx = k0 + (uint32_t)lane;
                    int p_l = m_idx + 1 < off_n ? S0 + (int)OFF[m_idx] : ip_limit + 16;
                    int inc = m_idx + 1 < off_n ? (int)(OFF[m_idx + 1] - OFF[m_idx]) : 1;
                    bool valid = p_l + inc <= ip_limit;
                    uint32_t v_l = (p_l >= 0 && p_l + 4 <= (int)n) ? rd32g(p_l) : 0;
                    uint32_t h_l = snp_hash(v_l, shift);
                    uint16_t t_l = s_table[h_l];
                    uint32_t spec_cand = valid ? rd32g((int)t_l) : 0;
                    // in-window duplicate groups via hash-bit ballots, all in
                    // registers (same transform as the LZ4 kernel): no LDS
                    // marker round trip, no table clobber/restore. hash_bits
                    // = 32 - shift (snappy's table is input-size dependent).
                    int pred = -1;
                    {
                        uint64_t mem = ~0ull;
                        for (int j = 0; j < 32 - shift; j++) {
                            uint64_t Bj = __ballot(((h_l >> j) & 1u) != 0);
                            mem &= ((h_l >> j) & 1u) ? Bj : ~Bj;
                        }
                        uint64_t below = mem & ((1ULL << lane) - 1);
                        if (below) pred = 63 - (int)__clzll((long long)below);
                    }
                    int pred_idx = pred >= 0 ? pred : 0;
                    int pred_pos = __shfl(p_l, pred_idx);
                    uint32_t pred_val = (uint32_t)__shfl((int)v_l, pred_idx);
                    int cand_pos = pred >= 0 ? pred_pos : (int)t_l;
                    uint32_t cand_val = pred >= 0 ? pred_val : spec_cand;
                    bool m_l = valid && cand_val == v_l;
                    uint64_t abort_mask = __ballot(!valid);
                    int first_abort = abort_mask ? (int)__ffsll((long long)abort_mask) - 1 : WAVE;
                    uint64_t match_mask = __ballot(m_l);
                    int first_event = match_mask ? (int)__ffsll((long long)match_mask) - 1 : WAVE;
                    bool have_match = first_event < first_abort && first_event < WAVE;
                    int commit_hi = have_match ? first_event
                                               : (first_abort < WAVE ? first_abort - 1 : WAVE - 1);
                    // commit only (the table was never clobbered): conflicting
                    // same-slot stores retire highest-lane-last == the scalar
                    // loop's last-write-wins order
                    if (lane <= commit_hi) {
                        volatile uint16_t* vt = s_table;
                        vt[h_l] = (uint16_t)p_l;
                    }
                    if (have_match) {
                        ip = __shfl(p_l, first_event);
                        match = __shfl(cand_pos, first_event);
                        found = true;
                    } else if (first_abort < WAVE) {
                        aborted = true;
                    }
                    if (found || aborted) break;
                    k0 += WAVE;
                }
                if (aborted) break;  // -> emit remainder
            }
            // ---- literal ----
            emit_literal(next_emit, ip - next_emit);
            // ---- copy loop (uniform) ----
            {
                while (true) {
                    int base = ip;
                    // FindMatchLength(match+4, ip+4, n): dword-per-lane; a
                    // byte at/after n is a mismatch (same bound semantics)
                    int mc = 0;
                    {
                        int offb = 0;
                        while (true) {
                            int pi = ip + 4 + offb + 4 * lane;
                            int navail = (int)n - pi;
                            uint32_t aa = 0, bb = 0;
                            if (navail > 0) {
                                memcpy(&aa, s + pi, navail >= 4 ? 4u : (size_t)navail);
                                memcpy(&bb, s + match + 4 + offb + 4 * lane,
                                       navail >= 4 ? 4u : (size_t)navail);
                            }
                            uint32_t x = aa ^ bb;
                            int eq4 = navail <= 0 ? 0 : (x == 0 ? 4 : (__ffs((int)x) - 1) >> 3);
                            if (eq4 > navail) eq4 = navail;
                            uint64_t ne = __ballot(eq4 < 4);
                            if (ne) {
                                int fl = (int)__ffsll((long long)ne) - 1;
                                mc = offb + 4 * fl + __shfl(eq4, fl);
                                break;
                            }
                            offb += 4 * WAVE;
                        }
                    }
                    int matched = 4 + mc;
                    ip += matched;
                    emit_copy((uint32_t)(base - match), matched);
                    next_emit = ip;
                    if (ip >= ip_limit) goto emit_remainder;
                    uint64_t input_bytes;
                    memcpy(&input_bytes, s + ip - 1, 8);
                    uint32_t prev_hash = snp_hash((uint32_t)input_bytes, shift);
                    s_table[prev_hash] = (uint16_t)(ip - 1);
                    uint32_t cur_hash = snp_hash((uint32_t)(input_bytes >> 8), shift);
                    int candidate = s_table[cur_hash];
                    uint32_t candidate_bytes = rd32g(candidate);
                    s_table[cur_hash] = (uint16_t)ip;
                    if ((uint32_t)(input_bytes >> 8) != candidate_bytes) break;
                    match = candidate;
                }
                ip++;
            }
        }
    }
emit_remainder:
    if (next_emit < (int)n)
        emit_literal(next_emit, (int)n - next_emit);
    return (int)op;
}

// one wave per chunk; chunks are (src, n, dst, csize-out) quads
struct SnpChunk {
    const uint8_t* src;
    uint8_t* dst;
    uint32_t n;
};

__global__ void __launch_bounds__(WAVE) k_snappy_compress_wave(const SnpChunk* chunks,
                                                               uint32_t n_chunks,
                                                               uint32_t* csize,
                                                               const uint32_t* OFF,
                                                               uint32_t off_n) {
    __shared__ uint16_t s_table[SNP_MAX_TABLE];
    uint32_t c = blockIdx.x;
    if (c >= n_chunks) return;
    int lane = threadIdx.x;
    SnpChunk ch = chunks[c];
    uint32_t ts = snp_table_size(ch.n);
    int sz = snp_wave_compress(ch.src, ch.n, ch.dst, s_table, ts, OFF, off_n, lane);
    if (lane == 0) csize[c] = (uint32_t)sz;
}

// wave-cooperative snappy block decompressor: uniform tag chain on all
// lanes, wave-parallel copies (format_description.txt tag semantics)
__global__ void __launch_bounds__(WAVE) k_snappy_decompress_wave(const SnpChunk* chunks,
                                                                 uint32_t n_chunks,
                                                                 unsigned long long* error,
                                                                 uint8_t* bad_chunks) {
    uint32_t c = blockIdx.x;
    if (c >= n_chunks) return;
    int lane = threadIdx.x;
    SnpChunk ch = chunks[c];                // src = compressed, n = comp_len
    const uint8_t* in = ch.src;
    uint8_t* out = ch.dst;
    uint32_t ip = 0, iend = ch.n;
    auto fail = [&](unsigned long long code) {
        if (lane == 0) {
            if (bad_chunks) bad_chunks[c] = 1;
            else atomicExch(error, code);
        }
    };
    // varint uncompressed length
    uint32_t olen = 0;
    {
        int sh = 0;
        while (true) {
            if (ip >= iend || sh > 28) { fail(20); return; }
            uint8_t b = in[ip++];
            olen |= (uint32_t)(b & 0x7f) << sh;
            if (!(b & 0x80)) break;
            sh += 7;
        }
    }
    uint32_t opos = 0;
    while (ip < iend) {
        uint8_t tag = in[ip++];
        uint32_t kind = tag & 3;
        if (kind == 0) {                    // literal
            uint32_t len = (tag >> 2) + 1;
            if (len > 60) {
                uint32_t nb = len - 60;
                if (ip + nb > iend) { fail(21); return; }
                len = 0;
                for (uint32_t j = 0; j < nb; j++) len |= (uint32_t)in[ip + j] << (8 * j);
                len += 1;
                ip += nb;
            }
            if (ip + len > iend || opos + len > olen) { fail(22); return; }
            for (uint32_t i = 4 * (uint32_t)lane; i < len; i += 4 * WAVE) {
                uint32_t nb = len - i;
                if (nb >= 4) {
                    uint32_t v;
                    memcpy(&v, in + ip + i, 4);
                    memcpy(out + opos + i, &v, 4);
                } else {
                    for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = in[ip + i + j];
                }
            }
            ip += len;
            opos += len;
        } else {                            // copy
            uint32_t len, offset;
            if (kind == 1) {
                len = ((tag >> 2) & 7) + 4;
                if (ip >= iend) { fail(23); return; }
                offset = ((uint32_t)(tag >> 5) << 8) | in[ip++];
            } else if (kind == 2) {
                len = (tag >> 2) + 1;
                if (ip + 2 > iend) { fail(23); return; }
                offset = in[ip] | ((uint32_t)in[ip + 1] << 8);
                ip += 2;
            } else {                        // 4-byte offset (never emitted for 16K chunks)
                len = (tag >> 2) + 1;
                if (ip + 4 > iend) { fail(23); return; }
                memcpy(&offset, in + ip, 4);
                ip += 4;
            }
            if (offset == 0 || offset > opos || opos + len > olen) { fail(24); return; }
            const uint8_t* src = out + opos - offset;
            if (offset == 1) {
                uint32_t b4 = 0x01010101u * src[0];
                for (uint32_t i = 4 * (uint32_t)lane; i < len; i += 4 * WAVE) {
                    uint32_t nb = len - i;
                    if (nb >= 4) memcpy(out + opos + i, &b4, 4);
                    else for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = (uint8_t)b4;
                }
            } else if (len <= offset) {
                for (uint32_t i = 4 * (uint32_t)lane; i < len; i += 4 * WAVE) {
                    uint32_t nb = len - i;
                    if (nb >= 4) {
                        uint32_t v;
                        memcpy(&v, src + i, 4);
                        memcpy(out + opos + i, &v, 4);
                    } else {
                        for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = src[i + j];
                    }
                }
            } else {
                for (uint32_t i = (uint32_t)lane; i < len; i += WAVE)
                    out[opos + i] = src[i % offset];
            }
            opos += len;
        }
    }
    if (opos != olen) fail(25);
}

// ---------------------------------------------------------------------------
// product-pipeline kernels: same slot/csize/ccrc contract as the LZ4 pair in
// lz4_wave.h (slot stride SNP_SLOT; payload = varint + fragment, CRC32 over
// the payload; CompressedSequentialWriter framing is codec-agnostic)
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(WAVE) k_snappy_compress_chunks(
    const uint8_t* data, uint64_t data_len, uint8_t* slots, uint32_t* csize, uint32_t* ccrc,
    uint32_t n_chunks, const uint32_t* crc_table, const uint32_t* OFF, uint32_t off_n) {
    __shared__ uint16_t s_table[SNP_MAX_TABLE];
    uint32_t c = blockIdx.x;
    if (c >= n_chunks) return;
    int lane = threadIdx.x;
    uint64_t off = (uint64_t)c * CHUNK_LEN;
    uint32_t len = (uint32_t)min((uint64_t)CHUNK_LEN, data_len - off);
    uint8_t* dst = slots + (uint64_t)c * SNP_SLOT;
    int csz = snp_wave_compress(data + off, len, dst, s_table, snp_table_size(len), OFF, off_n,
                                lane);
    if (lane == 0) {
        csize[c] = (uint32_t)csz;
        // slice-by-8 (crc_table is the 8x256 sliced set): 8 independent
        // table hits per serial round instead of one
        uint32_t crc = 0xFFFFFFFFu;
        int i = 0;
        for (; i + 8 <= csz; i += 8) {
            uint32_t lo, hi;
            memcpy(&lo, dst + i, 4);
            memcpy(&hi, dst + i + 4, 4);
            lo ^= crc;
            crc = crc_table[7 * 256 + (lo & 0xFF)] ^ crc_table[6 * 256 + ((lo >> 8) & 0xFF)] ^
                  crc_table[5 * 256 + ((lo >> 16) & 0xFF)] ^ crc_table[4 * 256 + (lo >> 24)] ^
                  crc_table[3 * 256 + (hi & 0xFF)] ^ crc_table[2 * 256 + ((hi >> 8) & 0xFF)] ^
                  crc_table[1 * 256 + ((hi >> 16) & 0xFF)] ^ crc_table[0 * 256 + (hi >> 24)];
        }
        for (; i < csz; i++) crc = crc_table[(crc ^ dst[i]) & 0xFF] ^ (crc >> 8);
        ccrc[c] = ~crc;
    }
}

// decompress on the ingest ChunkDesc contract (mirror of k_lz4_decompress_wave
// incl. the trailing per-lane CRC blocks; grid via lz4_decomp_grid)
__global__ void __launch_bounds__(WAVE) k_snappy_decompress_chunks(const ChunkDesc* chunks,
                                                                   uint32_t n, int verify_crc,
                                                                   unsigned long long* error,
                                                                   const uint32_t* crc_table,
                                                                   uint8_t* bad_chunks = nullptr) {
    // grid shared with the LZ4 decoder (lz4_decomp_grid): TWO chunks per
    // wave (one per half-wave — overlaps the serial tag-chain latencies),
    // CRC verification in trailing one-chunk-per-LANE blocks
    const uint32_t DB = lz4_decomp_blocks(n);
    uint32_t c = blockIdx.x * 2 + (threadIdx.x >> 5);
    int lane = threadIdx.x & 31;
    constexpr int HW = 32;
    if (blockIdx.x >= DB) {
        uint32_t ci = (blockIdx.x - DB) * WAVE + (uint32_t)threadIdx.x;
        if (ci >= n) return;
        ChunkDesc ch = chunks[ci];
        if (ch.comp_len > SNP_SLOT) return;  // decode block flags it
        const uint8_t* s_comp = ch.comp;
        uint32_t crc = 0xFFFFFFFFu;
        uint32_t i = 0;
        for (; i + 8 <= ch.comp_len; i += 8) {
            uint32_t lo, hi;
            memcpy(&lo, s_comp + i, 4);
            memcpy(&hi, s_comp + i + 4, 4);
            lo ^= crc;
            crc = crc_table[7 * 256 + (lo & 0xFF)] ^ crc_table[6 * 256 + ((lo >> 8) & 0xFF)] ^
                  crc_table[5 * 256 + ((lo >> 16) & 0xFF)] ^ crc_table[4 * 256 + (lo >> 24)] ^
                  crc_table[3 * 256 + (hi & 0xFF)] ^ crc_table[2 * 256 + ((hi >> 8) & 0xFF)] ^
                  crc_table[1 * 256 + ((hi >> 16) & 0xFF)] ^ crc_table[0 * 256 + (hi >> 24)];
        }
        for (; i < ch.comp_len; i++) crc = crc_table[(crc ^ s_comp[i]) & 0xFF] ^ (crc >> 8);
        crc = ~crc;
        uint32_t stored = ((uint32_t)s_comp[ch.comp_len] << 24) |
                          ((uint32_t)s_comp[ch.comp_len + 1] << 16) |
                          ((uint32_t)s_comp[ch.comp_len + 2] << 8) | s_comp[ch.comp_len + 3];
        if (crc != stored) { if (bad_chunks) bad_chunks[ci] = 1; else atomicExch(error, 1ull); }
        return;
    }
    (void)verify_crc;
    if (c >= n) return;  // odd tail half-wave
    ChunkDesc ch = chunks[c];
    if (ch.comp_len > SNP_SLOT) { if (lane == 0) { if (bad_chunks) bad_chunks[c] = 1; else atomicExch(error, 9ull); } return; }
    const uint8_t* in = ch.comp;
    uint8_t* out = ch.out;
    uint32_t ip = 0, iend = ch.comp_len, olen = ch.out_len;
    auto fail = [&](unsigned long long code) {
        if (lane == 0) {
            if (bad_chunks) bad_chunks[c] = 1;
            else atomicExch(error, code);
        }
    };
    uint32_t hdr = 0;
    {
        int sh = 0;
        while (true) {
            if (ip >= iend || sh > 28) { fail(2); return; }
            uint8_t b = in[ip++];
            hdr |= (uint32_t)(b & 0x7f) << sh;
            if (!(b & 0x80)) break;
            sh += 7;
        }
    }
    if (hdr != olen) { fail(2); return; }
    uint32_t opos = 0;
    while (ip < iend) {
        uint8_t tag = in[ip++];
        uint32_t kind = tag & 3;
        if (kind == 0) {
            uint32_t len = (tag >> 2) + 1;
            if (len > 60) {
                uint32_t nb = len - 60;
                if (ip + nb > iend) { fail(3); return; }
                len = 0;
                for (uint32_t j = 0; j < nb; j++) len |= (uint32_t)in[ip + j] << (8 * j);
                len += 1;
                ip += nb;
            }
            if (ip + len > iend || opos + len > olen) { fail(3); return; }
            for (uint32_t i = 4 * (uint32_t)lane; i < len; i += 4 * HW) {
                uint32_t nb = len - i;
                if (nb >= 4) {
                    uint32_t v;
                    memcpy(&v, in + ip + i, 4);
                    memcpy(out + opos + i, &v, 4);
                } else {
                    for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = in[ip + i + j];
                }
            }
            ip += len;
            opos += len;
        } else {
            uint32_t len, offset;
            if (kind == 1) {
                len = ((tag >> 2) & 7) + 4;
                if (ip >= iend) { fail(4); return; }
                offset = ((uint32_t)(tag >> 5) << 8) | in[ip++];
            } else if (kind == 2) {
                len = (tag >> 2) + 1;
                if (ip + 2 > iend) { fail(4); return; }
                offset = in[ip] | ((uint32_t)in[ip + 1] << 8);
                ip += 2;
            } else {
                len = (tag >> 2) + 1;
                if (ip + 4 > iend) { fail(4); return; }
                memcpy(&offset, in + ip, 4);
                ip += 4;
            }
            if (offset == 0 || offset > opos || opos + len > olen) { fail(4); return; }
            const uint8_t* src = out + opos - offset;
            if (offset == 1) {
                uint32_t b4 = 0x01010101u * src[0];
                for (uint32_t i = 4 * (uint32_t)lane; i < len; i += 4 * HW) {
                    uint32_t nb = len - i;
                    if (nb >= 4) memcpy(out + opos + i, &b4, 4);
                    else for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = (uint8_t)b4;
                }
            } else if (len <= offset) {
                for (uint32_t i = 4 * (uint32_t)lane; i < len; i += 4 * HW) {
                    uint32_t nb = len - i;
                    if (nb >= 4) {
                        uint32_t v;
                        memcpy(&v, src + i, 4);
                        memcpy(out + opos + i, &v, 4);
                    } else {
                        for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = src[i + j];
                    }
                }
            } else {
                for (uint32_t i = (uint32_t)lane; i < len; i += HW)
                    out[opos + i] = src[i % offset];
            }
            opos += len;
        }
    }
    if (opos != olen) fail(5);
}

}  // namespace gpuc
