// CPU simulator of the wave LZ4 compressor (shared by sim fuzz + GPU debug).
#pragma once
#include "../../cassandra_amd/csrc/lz4_model.h"
#include <cstdint>
#include <cstring>
#define WAVE 64
static int g_adv_g(int T) {
    if (T < 0) return 0;
    int S = T >> 6;
    return 64 * (S * (S - 1) / 2) + S * (T - 64 * S + 1);
}
static int g_adv_f(int x) { return g_adv_g(63 + x) - g_adv_g(63); }
static int adv_sum(int a, int b) {
    if (b <= a) return 0;
    int d = 0;
    if (a == 0) { d += 1; a = 1; if (b <= a) return d; }
    return d + g_adv_f(b - 1) - g_adv_f(a - 1);
}
static int adv1(int m) { return m == 0 ? 1 : (63 + m) >> 6; }
static uint32_t rd32(const uint8_t* s, int p) { uint32_t v; memcpy(&v, s + p, 4); return v; }

// returns compressed size; mirrors lz4_wave_compress control flow
#include <vector>
static int sim_compress(const uint8_t* s, int srcSize, uint8_t* dst, uint16_t* table,
                        std::vector<uint32_t>* eps = nullptr) {
    const int mflimitPlusOne = srcSize - LZ4M_MFLIMIT + 1;
    const int matchlimit = srcSize - LZ4M_LASTLITERALS;
    int ip = 0, anchor = 0;
    int op = 0;
    auto last_literals = [&]() {
        int lastRun = srcSize - anchor;
        if (lastRun >= (int)LZ4M_RUN_MASK) {
            int acc = lastRun - LZ4M_RUN_MASK;
            dst[op++] = (uint8_t)(LZ4M_RUN_MASK << LZ4M_ML_BITS);
            for (; acc >= 255; acc -= 255) dst[op++] = 255;
            dst[op++] = (uint8_t)acc;
        } else dst[op++] = (uint8_t)(lastRun << LZ4M_ML_BITS);
        memcpy(dst + op, s + anchor, srcSize - anchor);
        op += srcSize - anchor;
    };
    if (srcSize < LZ4M_MFLIMIT + 1) { last_literals(); return op; }
    table[lz4m_hash(rd32(s, 0))] = 0;
    ip = 1;
    bool done = false;
    while (!done) {
        int match = -1;
        {
            int k0 = 0, P0 = ip;
            bool found = false, aborted = false;
            while (true) {
                int p[WAVE], pred[WAVE], maxgroup[WAVE];
                uint32_t v[WAVE], h[WAVE];
                uint16_t t[WAVE];
                bool valid[WAVE], m[WAVE];
                for (int l = 0; l < WAVE; l++) {
                    p[l] = P0 + adv_sum(k0, k0 + l);
                    valid[l] = (p[l] + adv1(k0 + l)) <= mflimitPlusOne;
                    v[l] = (p[l] + 4 <= srcSize && p[l] >= 0) ? rd32(s, p[l]) : 0;
                    h[l] = lz4m_hash(v[l]);
                    t[l] = table[h[l]];
                }
                // highest-lane-wins marker write + readback
                for (int l = 0; l < WAVE; l++) table[h[l]] = (uint16_t)l;
                for (int l = 0; l < WAVE; l++) maxgroup[l] = table[h[l]];
                for (int l = 0; l < WAVE; l++) pred[l] = -1;
                {
                    uint64_t G = 0;
                    for (int l = 0; l < WAVE; l++) if (maxgroup[l] != l) G |= 1ULL << l;
                    while (G) {
                        int g = __builtin_ctzll(G);
                        int mg = maxgroup[g];
                        uint64_t members = 0;
                        for (int l = 0; l < WAVE; l++) if (maxgroup[l] == mg) members |= 1ULL << l;
                        for (int l = 0; l < WAVE; l++)
                            if (maxgroup[l] == mg) {
                                uint64_t below = members & ((1ULL << l) - 1);
                                if (below) pred[l] = 63 - __builtin_clzll(below);
                            }
                        G &= ~members;
                    }
                }
                for (int l = 0; l < WAVE; l++) {
                    if (pred[l] >= 0) m[l] = valid[l] && v[pred[l]] == v[l];
                    else m[l] = valid[l] && rd32(s, t[l]) == v[l];
                }
                int first_abort = WAVE, first_event = WAVE;
                for (int l = 0; l < WAVE; l++) if (!valid[l]) { first_abort = l; break; }
                for (int l = 0; l < WAVE; l++) if (m[l]) { first_event = l; break; }
                bool have_match = first_event < first_abort && first_event < WAVE;
                int commit_hi = have_match ? first_event : (first_abort < WAVE ? first_abort - 1 : WAVE - 1);
                for (int l = 0; l < WAVE; l++) table[h[l]] = t[l];
                for (int l = 0; l < WAVE; l++) if (l <= commit_hi) table[h[l]] = (uint16_t)p[l];
                if (have_match) {
                    ip = p[first_event];
                    match = pred[first_event] >= 0 ? p[pred[first_event]] : (int)t[first_event];
                    found = true;
                } else if (first_abort < WAVE) aborted = true;
                if (found || aborted) break;
                P0 += adv_sum(k0, k0 + WAVE);
                k0 += WAVE;
            }
            if (aborted) { last_literals(); return op; }
        }
        // catch up
        while (ip > anchor && match > 0 && s[ip - 1] == s[match - 1]) { ip--; match--; }
        // literals
        {
            int lit = ip - anchor;
            int token_pos = op++;
            uint8_t token;
            if (lit >= (int)LZ4M_RUN_MASK) {
                token = (uint8_t)(LZ4M_RUN_MASK << LZ4M_ML_BITS);
                int len = lit - LZ4M_RUN_MASK;
                for (; len >= 255; len -= 255) dst[op++] = 255;
                dst[op++] = (uint8_t)len;
            } else token = (uint8_t)(lit << LZ4M_ML_BITS);
            memcpy(dst + op, s + anchor, lit);
            op += lit;
            while (true) {
                int offv = ip - match;
                if (eps) { eps->push_back((uint32_t)ip); eps->push_back((uint32_t)match); eps->push_back((uint32_t)lit); }
                dst[op] = (uint8_t)offv;
                dst[op + 1] = (uint8_t)(offv >> 8);
                op += 2;
                int mc = 0;
                while (ip + 4 + mc < matchlimit && s[ip + 4 + mc] == s[match + 4 + mc]) mc++;
                ip += mc + LZ4M_MINMATCH;
                if (mc >= (int)LZ4M_ML_MASK) {
                    token += LZ4M_ML_MASK;
                    int rem = mc - LZ4M_ML_MASK;
                    for (; rem >= 255; rem -= 255) dst[op++] = 255;
                    dst[op++] = (uint8_t)rem;
                } else token += (uint8_t)mc;
                dst[token_pos] = token;
                anchor = ip;
                if (ip >= mflimitPlusOne) { done = true; break; }
                table[lz4m_hash(rd32(s, ip - 2))] = (uint16_t)(ip - 2);
                uint32_t vv = rd32(s, ip);
                uint32_t hh = lz4m_hash(vv);
                int m2 = (int)table[hh];
                table[hh] = (uint16_t)ip;
                if (rd32(s, m2) == vv) {
                    match = m2;
                    token_pos = op++;
                    token = 0;
                    continue;
                }
                ip += 1;
                break;
            }
            if (done) break;
        }
    }
    last_literals();
    return op;
}

