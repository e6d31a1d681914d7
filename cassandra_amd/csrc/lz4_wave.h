// PRODUCT — wave-cooperative LZ4 kernels for CDNA4 (gfx950).
//
// Bit-exact reimplementation of LZ4_compress_default (liblz4 1.9.3, byU16,
// acceleration 1 — the scalar spec lives in lz4_model.h and is fuzz-pinned to
// the system library) where the sequential hash-probe chain is executed as
// 64-position speculative windows across the wavefront:
//
//   * the probe sequence advances by adv_0=1, adv_m=(63+m)>>6 (the scalar
//     loop's skip acceleration); a wave evaluates 64 probes per step at the
//     exact closed-form positions (lz4_adv_sum), tests candidates in
//     parallel, then commits table updates only up to the first match/abort
//     — byte-identical to the scalar loop.
//   * in-window hash duplicates are resolved exactly: one marker write/read
//     round on the LDS hash table names each hash group's max lane (CDNA LDS
//     conflict rule: simultaneous writes to one address retire in lane order,
//     highest lane wins — k_probe_lds_order verifies this at init and the
//     host refuses to run if it does not hold), then a per-group ballot walk
//     hands every lane its exact in-window predecessor.
//   * match extension, catch-up, literal copies and 255-run emission are
//     wave-parallel (ballots + strided copies).
//
// Decompression runs the token chain redundantly on all 64 lanes (uniform
// control flow; operands come from LDS broadcasts) with wave-parallel copies;
// overlapped match copies use the periodic form dst[i] = src[i % offset].
#pragma once
#include <hip/hip_runtime.h>
#include "codec.h"
#include "lz4_model.h"

namespace gpuc {

#ifndef WAVE
#define WAVE 64
#endif

__device__ inline uint64_t wave_ballot(bool p) { return __ballot(p); }

// probe: do conflicting LDS writes retire highest-lane-last?
__global__ void k_probe_lds_order(unsigned int* out) {
    __shared__ unsigned int slot[1];
    __shared__ unsigned int slot2[1];
    int lane = threadIdx.x & (WAVE - 1);
    slot[0] = 0xFFFF;
    __syncthreads();
    slot[0] = (unsigned)lane;          // all 64 lanes, same address
    slot2[0] = (unsigned)(63 - lane);  // reversed
    __syncthreads();
    if (threadIdx.x == 0) { out[0] = slot[0]; out[1] = slot2[0]; }
}

// ---------------------------------------------------------------------------
// wave-cooperative compress: one wave per 16 KiB chunk
// s_chunk: staged input; s_table: u16[8192] zeroed; dst: global output (after
// the caller-written 4-byte LE header). Returns compressed size via lane 0.
// ---------------------------------------------------------------------------
__device__ inline uint32_t lds_read32(const uint8_t* s, uint32_t p) {
    uint32_t v;
    memcpy(&v, s + p, 4);
    return v;
}

// advance of the scalar probe loop after probe m (lz4.c: step initialised to 1,
// then step = searchMatchNb++ >> 6 with searchMatchNb starting at 64):
// adv_0 = 1, adv_m = (63+m)>>6 for m >= 1. lz4_adv_sum(a,b) = sum adv_m, m in [a,b).
__device__ inline int lz4_adv_g(int T) {  // sum_{t=0}^{T} t>>6
    if (T < 0) return 0;
    int S = T >> 6;
    return 64 * (S * (S - 1) / 2) + S * (T - 64 * S + 1);
}
__device__ inline int lz4_adv_f(int x) {  // sum_{m=1}^{x} (63+m)>>6
    return lz4_adv_g(63 + x) - lz4_adv_g(63);
}
__device__ inline int lz4_adv_sum(int a, int b) {
    if (b <= a) return 0;
    int d = 0;
    if (a == 0) {
        d += 1;
        a = 1;
        if (b <= a) return d;
    }
    return d + lz4_adv_f(b - 1) - lz4_adv_f(a - 1);
}
__device__ inline int lz4_adv(int m) { return m == 0 ? 1 : (63 + m) >> 6; }

// wave-parallel byte-exact copy, dword per lane (src/dst any alignment,
// regions must not overlap)
__device__ inline void wave_copy(uint8_t* __restrict__ dst, const uint8_t* __restrict__ src,
                                 int nbytes, int lane) {
    for (int i = 4 * lane; i < nbytes; i += 4 * WAVE) {
        int nb = nbytes - i;
        if (nb >= 4) {
            uint32_t v;
            memcpy(&v, src + i, 4);
            memcpy(dst + i, &v, 4);
        } else {
            for (int j = 0; j < nb; j++) dst[i + j] = src[i + j];
        }
    }
}

template <bool STAGE_LDS = true>
__device__ inline int lz4_wave_compress(const uint8_t* __restrict__ s_chunk, int srcSize,
                                        uint8_t* __restrict__ dst, uint16_t* __restrict__ s_table,
                                        int lane, uint32_t* dbg = nullptr) {
    const int mflimitPlusOne = srcSize - LZ4M_MFLIMIT + 1;
    const int matchlimit = srcSize - LZ4M_LASTLITERALS;
    int ip = 0, anchor = 0;
    uint32_t op = 0;

    auto emit_last_literals = [&]() {
        int lastRun = srcSize - anchor;
        if (lastRun >= (int)LZ4M_RUN_MASK) {
            int acc = lastRun - LZ4M_RUN_MASK;
            if (lane == 0) dst[op] = (uint8_t)(LZ4M_RUN_MASK << LZ4M_ML_BITS);
            op++;
            int n255 = acc / 255;
            for (int i = lane; i < n255; i += WAVE) dst[op + i] = 255;
            op += n255;
            if (lane == 0) dst[op] = (uint8_t)(acc % 255);
            op++;
        } else {
            if (lane == 0) dst[op] = (uint8_t)(lastRun << LZ4M_ML_BITS);
            op++;
        }
        wave_copy(dst + op, s_chunk + anchor, lastRun, lane);
        op += lastRun;
    };

    if (srcSize < LZ4M_MFLIMIT + 1) {
        emit_last_literals();
        return (int)op;
    }

    // first byte insert (all lanes uniform)
    s_table[lz4m_hash(lds_read32(s_chunk, 0))] = 0;
    ip = 1;

    // ---- match-extension resolver ----------------------------------------
    // Counts common bytes of s_chunk[ip+4..] vs s_chunk[match+4..] up to
    // matchlimit (dword per lane, 256 B per ballot round). Round-0 operands
    // may be preloaded by the caller so their global loads overlap the
    // catch-up round's. On return w8/w8v carry the 8 bytes at (ip+4)+(mc-2)
    // — the next table-fill dword (new_ip-2) and immediate-test dword
    // (new_ip) — reconstructed from the breaking round's registers when the
    // span lies inside lanes [fl-1, fl+1] of that round (else the caller
    // loads them; both only matter when the compressor continues, in which
    // case lane fl+1's dword is provably in range and loaded).
    uint64_t w8 = 0;
    bool w8v = false;
    auto ext_count = [&](int eip, int ematch, bool pre, uint32_t aa0, uint32_t bb0) -> int {
        int offb = 0;
        while (true) {
            int pi = eip + 4 + offb + 4 * lane;
            int navail = matchlimit - pi;
            uint32_t aa = 0, bb = 0;
            if (offb == 0 && pre) {
                aa = aa0;
                bb = bb0;
            } else if (navail > 0) {
                aa = lds_read32(s_chunk, (uint32_t)pi);
                bb = lds_read32(s_chunk, (uint32_t)(ematch + 4 + offb + 4 * lane));
            }
            uint32_t x = aa ^ bb;
            int eq4 = navail <= 0 ? 0 : (x == 0 ? 4 : (__ffs((int)x) - 1) >> 3);
            if (eq4 > navail) eq4 = navail;
            uint64_t ne = wave_ballot(eq4 < 4);
            if (ne) {
                int fl = (int)__ffsll((long long)ne) - 1;
                int eqf = __shfl(eq4, fl);
                // 8-byte window starting at round offset 4*fl + eqf - 2 spans
                // lanes fl-1 (eqf<2), fl, fl+1 (eqf>=1)
                w8v = !((eqf < 2 && fl == 0) || (eqf >= 1 && fl == 63));
                if (w8v) {
                    uint32_t A = (uint32_t)__shfl((int)aa, fl > 0 ? fl - 1 : 0);
                    uint32_t B = (uint32_t)__shfl((int)aa, fl);
                    uint32_t C = (uint32_t)__shfl((int)aa, fl < 63 ? fl + 1 : 63);
                    uint64_t AB = (uint64_t)A | ((uint64_t)B << 32);
                    uint64_t BC = (uint64_t)B | ((uint64_t)C << 32);
                    w8 = eqf >= 2 ? (BC >> (8 * (eqf - 2)))
                                  : ((AB >> (8 * (eqf + 2))) | ((uint64_t)C << (8 * (6 - eqf))));
                }
                return offb + 4 * fl + eqf;
            }
            offb += 4 * WAVE;
        }
    };

    // cross-sequence prefetch: the sequence tail issues the next window's
    // round-0 probe loads before its immediate-test chain resolves (pure
    // loads at closed-form positions — semantics unchanged)
    int pre_p = -1;
    uint32_t pre_v = 0;
    bool pre_ok = false;
    bool done = false;
    while (!done) {
        // ================= match finder: 64-probe windows =================
        int match = -1;           // matched candidate position
        {
            int k0 = 0;           // probe index of this window's lane 0
            int P0 = ip;          // its position
            bool found = false, aborted = false;
            // round-0 probe dwords: taken from the sequence tail's prefetch
            // when it targeted this ip; inside the loop the NEXT window's
            // dwords are prefetched while the current one resolves
            int p_l = P0 + lz4_adv_sum(k0, k0 + lane);
            uint32_t v_l;
            if (pre_ok && pre_p == p_l) {
                v_l = pre_v;
            } else {
                v_l = (p_l >= 0 && p_l + 4 <= srcSize) ? lds_read32(s_chunk, (uint32_t)p_l) : 0;
            }
            pre_ok = false;
            while (true) {
                // scalar loop aborts probe m when q_m + adv_m > mflimitPlusOne
                bool valid = (p_l + lz4_adv(k0 + lane)) <= mflimitPlusOne;
                uint32_t h_l = lz4m_hash(v_l);
                uint16_t t_l = s_table[h_l];              // pre-window candidate
                // speculative candidate dword for the (common) no-duplicate
                // case: issue the load before the ballot section. Loads only
                // on `valid` lanes: an invalid lane's (in-window predecessor)
                // position can lie past the chunk end — the scalar code never
                // evaluates those candidates at all.
                uint32_t spec_cand = valid ? lds_read32(s_chunk, (uint32_t)t_l) : 0;
                // prefetch next window probes
                int p_n = P0 + lz4_adv_sum(k0, k0 + WAVE + lane);
                uint32_t v_n = (p_n >= 0 && p_n + 4 <= srcSize) ? lds_read32(s_chunk, (uint32_t)p_n) : 0;
                // in-window duplicate groups, all in registers: 13 ballots
                // over the (HASHLOG+1)-bit hash give each lane the exact set
                // of lanes probing the same table slot — no LDS marker round
                // trip, no table clobber/restore, and the ballots overlap the
                // spec_cand load latency. A lane's in-window predecessor (the
                // scalar loop's latest earlier probe with this hash) is the
                // highest same-hash lane strictly below it; invalid lanes are
                // a suffix (positions/advances are monotone in lane), so a
                // valid lane's predecessor is always a valid lane.
                uint64_t mem = ~0ull;
#pragma unroll
                for (int j = 0; j < LZ4M_HASHLOG + 1; j++) {
                    uint64_t Bj = wave_ballot(((h_l >> j) & 1u) != 0);
                    mem &= ((h_l >> j) & 1u) ? Bj : ~Bj;
                }
                uint64_t below = mem & ((1ULL << lane) - 1);
                int pred = below ? 63 - (int)__clzll((long long)below) : -1;
                // exact match test: in-window predecessor position if any,
                // else the pre-window table entry. The predecessor lane
                // already holds both its probe position AND its probe dword
                // in registers — shfl them instead of re-deriving the
                // position and re-loading the dword.
                int pred_idx = pred >= 0 ? pred : 0;
                uint32_t pred_pos = (uint32_t)__shfl(p_l, pred_idx);
                uint32_t pred_val = (uint32_t)__shfl((int)v_l, pred_idx);
                uint32_t cand_pos = pred >= 0 ? pred_pos : (uint32_t)t_l;
                uint32_t cand_val = pred >= 0 ? pred_val : spec_cand;
                bool m_l = valid && cand_val == v_l;
                uint64_t abort_mask = wave_ballot(!valid);
                int first_abort = abort_mask ? (int)__ffsll((long long)abort_mask) - 1 : WAVE;
                uint64_t match_mask = wave_ballot(m_l);
                int first_event = match_mask ? (int)__ffsll((long long)match_mask) - 1 : WAVE;
                bool have_match = first_event < first_abort && first_event < WAVE;
                int commit_hi = have_match ? first_event : (first_abort < WAVE ? first_abort - 1 : WAVE - 1);
                // commit the probed range: only lanes up to the match/abort
                // point write (the table was never clobbered, so no restore
                // pass). Conflicting same-slot writes retire highest-lane-
                // last (k_probe_lds_order verifies at init), matching the
                // scalar loop's last-write-wins order.
                if (lane <= commit_hi) {
                    volatile uint16_t* vt = s_table;
                    vt[h_l] = (uint16_t)p_l;
                }
                if (have_match) {
                    ip = __shfl(p_l, first_event);  // == P0 + adv_sum(k0, k0+first_event)
                    match = (int)(uint32_t)__shfl((int)cand_pos, first_event);
                    found = true;
                } else if (first_abort < WAVE) {
                    aborted = true;
                }
                if (found || aborted) break;
                P0 += lz4_adv_sum(k0, k0 + WAVE);
                k0 += WAVE;
                p_l = p_n;
                v_l = v_n;
            }
            if (aborted) { emit_last_literals(); return (int)op; }
        }

        // ========== catch-up ∥ match extension (round-0 loads fused) ======
        // The scalar loop extends the match backward (catch-up) and then
        // forward from the moved ip. Forward extension from the PRE-catch-up
        // ip compares the same byte pairs shifted by `run`, and the first
        // `run` post-catch-up pairs are already known equal (they lie in the
        // catch-up run / verified match dword), so
        //   mc_post == run + mc_pre        (same matchlimit clamp)
        // — both rounds' loads issue together and resolve in one global
        // round trip instead of two.
        int mc;
        {
            int e = lane + 1;
            bool cu_ok = (ip - e + 1 > anchor) && (match - e + 1 > 0) &&
                         s_chunk[ip - e] == s_chunk[match - e];
            uint32_t aa0 = 0, bb0 = 0;
            {
                int pi0 = ip + 4 + 4 * lane;
                if (matchlimit - pi0 > 0) {
                    aa0 = lds_read32(s_chunk, (uint32_t)pi0);
                    bb0 = lds_read32(s_chunk, (uint32_t)(match + 4 + 4 * lane));
                }
            }
            uint64_t bad = wave_ballot(!cu_ok);
            int run = bad ? (int)__ffsll((long long)bad) - 1 : WAVE;
            {   // full-wave catch-up: keep walking backward (rare)
                int cip = ip - run, cmatch = match - run, r = run;
                while (r == WAVE) {
                    bool ok = (cip - e + 1 > anchor) && (cmatch - e + 1 > 0) &&
                              s_chunk[cip - e] == s_chunk[cmatch - e];
                    uint64_t bad2 = wave_ballot(!ok);
                    r = bad2 ? (int)__ffsll((long long)bad2) - 1 : WAVE;
                    run += r;
                    cip -= r;
                    cmatch -= r;
                }
            }
            mc = run + ext_count(ip, match, true, aa0, bb0);
            ip -= run;
            match -= run;
        }

        // ================= literals =================
        {
            const int lit = ip - anchor;
            uint32_t token_pos = op++;
            uint8_t token;
            if (lit >= (int)LZ4M_RUN_MASK) {
                token = (uint8_t)(LZ4M_RUN_MASK << LZ4M_ML_BITS);
                int len = lit - LZ4M_RUN_MASK;
                int n255 = len / 255;
                for (int i = lane; i < n255; i += WAVE) dst[op + i] = 255;
                op += n255;
                if (lane == 0) dst[op] = (uint8_t)(len % 255);
                op++;
            } else {
                token = (uint8_t)(lit << LZ4M_ML_BITS);
            }
            wave_copy(dst + op, s_chunk + anchor, lit, lane);
            op += lit;

            // ================= offset + match length =================
            // `mc` for the first iteration was computed above (fused with
            // catch-up); the immediate-match continue path computes its own.
            while (true) {
                int offv = ip - match;
                if (dbg && lane == 0) {
                    uint32_t n = ++dbg[0];
                    if (n < 2000) { dbg[n * 3] = (uint32_t)ip; dbg[n * 3 + 1] = (uint32_t)match; dbg[n * 3 + 2] = (uint32_t)lit; }
                }
                if (lane == 0) { dst[op] = (uint8_t)offv; dst[op + 1] = (uint8_t)(offv >> 8); }
                op += 2;
                ip += mc + LZ4M_MINMATCH;
                if (mc >= (int)LZ4M_ML_MASK) {
                    token += LZ4M_ML_MASK;
                    int rem = mc - LZ4M_ML_MASK;
                    int n255 = rem / 255;
                    for (int i = lane; i < n255; i += WAVE) dst[op + i] = 255;
                    op += n255;
                    if (lane == 0) dst[op] = (uint8_t)(rem % 255);
                    op++;
                } else {
                    token += (uint8_t)mc;
                }
                if (lane == 0) dst[token_pos] = token;

                anchor = ip;
                if (ip >= mflimitPlusOne) { done = true; break; }
                // prefetch the next window's round-0 probes (at ip+1, where
                // the finder restarts if the immediate test misses) so their
                // loads resolve under the immediate-test chain; discarded
                // harmlessly when the test hits
                pre_p = (ip + 1) + lz4_adv_sum(0, lane);
                pre_v = (pre_p + 4 <= srcSize) ? lds_read32(s_chunk, (uint32_t)pre_p) : 0;
                pre_ok = true;
                // table fill at ip-2 + immediate test at ip. Both dwords live
                // in w8 (extension registers, or one merged 8-byte load).
                // The scalar order is fill(ip-2) THEN read the test slot; do
                // the read first and patch it when the two hashes collide —
                // the final table state (both writes applied, test slot's ip
                // winning on collision) is unchanged.
                uint64_t w;
                if (w8v) {
                    w = w8;
                } else {
                    uint8_t const* q = s_chunk + (uint32_t)(ip - 2);
                    memcpy(&w, q, 8);
                }
                uint32_t v2 = (uint32_t)w;          // dword at ip-2
                uint32_t v = (uint32_t)(w >> 16);   // dword at ip
                uint32_t h = lz4m_hash(v);
                int m2 = (int)s_table[h];
                uint32_t h2 = lz4m_hash(v2);
                if (h2 == h) m2 = ip - 2;
                s_table[h2] = (uint16_t)(ip - 2);
                s_table[h] = (uint16_t)ip;
                if (lds_read32(s_chunk, (uint32_t)m2) == v) {
                    match = m2;
                    token_pos = op++;
                    token = 0;
                    mc = ext_count(ip, match, false, 0, 0);
                    continue;  // another match with empty literal run
                }
                ip += 1;
                break;
            }
            if (done) break;
        }
    }
    emit_last_literals();
    return (int)op;
}

template <bool STAGE_LDS>
__global__ void __launch_bounds__(WAVE) k_lz4_compress_wave_t(const uint8_t* data, uint64_t data_len,
                                                              uint8_t* slots, uint32_t* csize,
                                                              uint32_t* ccrc, uint32_t n_chunks,
                                                              const uint32_t* crc_table,
                                                              uint32_t* dbg = nullptr) {
    __shared__ uint16_t s_table[LZ4M_HASHTABLESIZE_U16];
    uint32_t c = blockIdx.x;
    if (c >= n_chunks) return;
    int lane = threadIdx.x;
    uint64_t off = (uint64_t)c * CHUNK_LEN;
    uint32_t len = (uint32_t)min((uint64_t)CHUNK_LEN, data_len - off);
    const uint8_t* src;
    if constexpr (STAGE_LDS) {
        __shared__ uint8_t s_chunk[CHUNK_LEN];
        for (uint32_t i = lane * 4; i + 4 <= len; i += WAVE * 4)
            *(uint32_t*)&s_chunk[i] = *(const uint32_t*)((const uint8_t*)data + off + i);
        for (uint32_t i = (len & ~3u) + lane; i < len; i += WAVE) s_chunk[i] = data[off + i];
        src = s_chunk;
    } else {
        src = data + off;
    }
    for (int i = lane; i < LZ4M_HASHTABLESIZE_U16; i += WAVE) s_table[i] = 0;
    __syncthreads();
    uint8_t* dst = slots + (uint64_t)c * LZ4_SLOT;
    if (lane == 0) {
        dst[0] = (uint8_t)len; dst[1] = (uint8_t)(len >> 8);
        dst[2] = (uint8_t)(len >> 16); dst[3] = (uint8_t)(len >> 24);
    }
    int csz = lz4_wave_compress(src, (int)len, dst + 4, s_table, lane, dbg);
    uint32_t total = (uint32_t)csz + 4;
    if (lane == 0) csize[c] = total;
    // CRC over the compressed bytes on lane 0, slice-by-8 (crc_table is the
    // 8x256 sliced set): 8 independent table hits per serial round instead
    // of one — ~8x fewer dependent round trips than the bytewise loop.
    if (lane == 0) {
        uint32_t crc = 0xFFFFFFFFu;
        uint32_t i = 0;
        for (; i + 8 <= total; i += 8) {
            uint32_t lo, hi;
            memcpy(&lo, dst + i, 4);
            memcpy(&hi, dst + i + 4, 4);
            lo ^= crc;
            crc = crc_table[7 * 256 + (lo & 0xFF)] ^ crc_table[6 * 256 + ((lo >> 8) & 0xFF)] ^
                  crc_table[5 * 256 + ((lo >> 16) & 0xFF)] ^ crc_table[4 * 256 + (lo >> 24)] ^
                  crc_table[3 * 256 + (hi & 0xFF)] ^ crc_table[2 * 256 + ((hi >> 8) & 0xFF)] ^
                  crc_table[1 * 256 + ((hi >> 16) & 0xFF)] ^ crc_table[0 * 256 + (hi >> 24)];
        }
        for (; i < total; i++) crc = crc_table[(crc ^ dst[i]) & 0xFF] ^ (crc >> 8);
        ccrc[c] = ~crc;
    }
}

// ---------------------------------------------------------------------------
// wave-cooperative decompress: comp chunk staged in LDS; the token chain runs
// redundantly on all lanes (uniform); copies are wave-parallel.
// ---------------------------------------------------------------------------
// grid helper: decode runs TWO chunks per wave (one per half-wave: the
// token chain is a serial pointer chase, so two divergent 32-lane decodes
// overlap their dependent-load latencies at no occupancy cost); with
// verify_crc the CRC work runs in extra trailing blocks (one chunk per
// LANE — the byte-serial CRC would otherwise hold all lanes of a decode
// block hostage behind one lane's loop)
__host__ __device__ static inline uint32_t lz4_decomp_blocks(uint32_t n) { return (n + 1) / 2; }
static inline uint32_t lz4_decomp_grid(uint32_t n, int verify_crc) {
    return lz4_decomp_blocks(n) + (verify_crc ? (n + WAVE - 1) / WAVE : 0);
}

__global__ void __launch_bounds__(WAVE) k_lz4_decompress_wave(const ChunkDesc* chunks, uint32_t n,
                                                              int verify_crc,
                                                              unsigned long long* error,
                                                              const uint32_t* crc_table,
                                                              uint8_t* bad_chunks = nullptr) {
    // compressed bytes are read through L1/L2 (all-lane same-address reads
    // broadcast; no LDS staging -> higher occupancy, same trade as the
    // compressor's global-src variant). crc_table is the 8x256 sliced set.
    const uint32_t DB = lz4_decomp_blocks(n);
    uint32_t c = blockIdx.x * 2 + (threadIdx.x >> 5);  // chunk per half-wave
    int lane = threadIdx.x & 31;                       // lane within the half
    constexpr int HW = 32;
    if (blockIdx.x >= DB) {
        // CRC block: lane l checks chunk (b-DB)*WAVE + l (launched only when
        // verify_crc; see lz4_decomp_grid). Each lane walks its own chunk
        // serially — consecutive iterations reuse the lane's cachelines.
        uint32_t ci = (blockIdx.x - DB) * WAVE + (uint32_t)threadIdx.x;
        if (ci >= n) return;
        ChunkDesc ch = chunks[ci];
        if (ch.comp_len > LZ4_SLOT) return;  // decode block flags it
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
        uint32_t stored = ((uint32_t)s_comp[ch.comp_len] << 24) | ((uint32_t)s_comp[ch.comp_len + 1] << 16) |
                          ((uint32_t)s_comp[ch.comp_len + 2] << 8) | s_comp[ch.comp_len + 3];
        if (crc != stored) { if (bad_chunks) bad_chunks[ci] = 1; else atomicExch(error, 1ull); }
        return;
    }
    if (c >= n) return;  // odd tail: second half-wave has no chunk
    ChunkDesc ch = chunks[c];
    if (ch.comp_len > LZ4_SLOT) { if (lane == 0) { if (bad_chunks) bad_chunks[c] = 1; else atomicExch(error, 9ull); } return; }
    const uint8_t* s_comp = ch.comp;
    uint32_t hdr = s_comp[0] | (s_comp[1] << 8) | (s_comp[2] << 16) | ((uint32_t)s_comp[3] << 24);
    if (hdr != ch.out_len) { if (lane == 0) { if (bad_chunks) bad_chunks[c] = 1; else atomicExch(error, 2ull); } return; }
    // uniform decode on all lanes
    uint32_t ip = 4, iend = ch.comp_len;
    uint32_t opos = 0, olen = ch.out_len;
    uint8_t* out = ch.out;
    while (ip < iend) {
        uint32_t token = s_comp[ip++];
        uint32_t lit = token >> 4;
        if (lit == 15) {
            uint32_t s;
            do { s = s_comp[ip++]; lit += s; } while (s == 255);
        }
        if (opos + lit > olen || ip + lit > iend) { if (lane == 0) { if (bad_chunks) bad_chunks[c] = 1; else atomicExch(error, 3ull); } return; }
        for (uint32_t i = 4u * lane; i < lit; i += 4u * HW) {
            uint32_t nb = lit - i;
            if (nb >= 4) {
                uint32_t v;
                memcpy(&v, s_comp + ip + i, 4);
                memcpy(out + opos + i, &v, 4);
            } else {
                for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = s_comp[ip + i + j];
            }
        }
        opos += lit; ip += lit;
        if (ip >= iend) break;
        uint32_t off = s_comp[ip] | (s_comp[ip + 1] << 8);
        ip += 2;
        uint32_t ml = (token & 15) + 4;
        if (ml == 19) {
            uint32_t s;
            do { s = s_comp[ip++]; ml += s; } while (s == 255);
        }
        if (off == 0 || opos < off || opos + ml > olen) { if (lane == 0) { if (bad_chunks) bad_chunks[c] = 1; else atomicExch(error, 4ull); } return; }
        const uint8_t* src = out + opos - off;
        if (off == 1) {
            // byte run: broadcast, dword stores
            uint32_t b4 = 0x01010101u * src[0];
            for (uint32_t i = 4u * lane; i < ml; i += 4u * HW) {
                uint32_t nb = ml - i;
                if (nb >= 4) memcpy(out + opos + i, &b4, 4);
                else for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = (uint8_t)b4;
            }
        } else if (ml <= off) {
            // disjoint regions: dword copy
            for (uint32_t i = 4u * lane; i < ml; i += 4u * HW) {
                uint32_t nb = ml - i;
                if (nb >= 4) {
                    uint32_t v;
                    memcpy(&v, src + i, 4);
                    memcpy(out + opos + i, &v, 4);
                } else {
                    for (uint32_t j = 0; j < nb; j++) out[opos + i + j] = src[i + j];
                }
            }
        } else {
            // overlapped copy (ml > off) == periodic repetition of the last
            // `off` bytes; the modulo form avoids reading bytes this same
            // copy has not written yet
            for (uint32_t i = lane; i < ml; i += HW) out[opos + i] = src[i % off];
        }
        opos += ml;
    }
    if (opos != olen && lane == 0) { if (bad_chunks) bad_chunks[c] = 1; else atomicExch(error, 5ull); }
}

}  // namespace gpuc
