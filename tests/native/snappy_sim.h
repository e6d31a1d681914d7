// CPU simulator of the planned wave-cooperative snappy compressor: executes
// the 64-probe-window decomposition (marker/predecessor machinery identical
// to the LZ4 kernel's) sequentially, so the decomposition can be proven
// byte-equal to snappy_model.h without a GPU. The HIP kernel is a direct
// transcription of this control flow.
#pragma once
#include "../../cassandra_amd/csrc/snappy_model.h"
#include <cstring>
#include <vector>

#define SNPW 64

// probe-offset table: OFF[m] = position of probe m relative to the run start
// (snappy.cc: skip=32; inc = skip>>5; skip += inc). OFF[m+1]-OFF[m] = inc_m.
struct SnpOff {
    std::vector<uint32_t> off;
    SnpOff() {
        uint32_t skip = 32, o = 0;
        off.push_back(0);
        while (o < (1u << 16) + 128) {
            uint32_t inc = skip >> 5;
            skip += inc;
            o += inc;
            off.push_back(o);
        }
    }
};
inline const SnpOff& snp_off() { static SnpOff t; return t; }

static int snp_sim_compress(const uint8_t* s, uint32_t n, uint8_t* dst, uint16_t* table,
                            uint32_t table_size) {
    using namespace gpuc;
    const auto& OFF = snp_off().off;
    uint8_t* op = dst;
    // varint length prefix
    {
        uint32_t v = n;
        while (v >= 0x80) { *op++ = (uint8_t)(v | 0x80); v >>= 7; }
        *op++ = (uint8_t)v;
    }
    memset(table, 0, table_size * sizeof(uint16_t));
    const int shift = 32 - __builtin_ctz(table_size);
    const int ip_limit = (int)n - (int)SNP_INPUT_MARGIN;  // index bound for next_ip
    int ip = 1;                                           // Hash(++ip) entry
    int next_emit = 0;
    auto rd32i = [&](int p) { return snp_load32(s + p); };

    if ((int)n >= (int)SNP_INPUT_MARGIN) {
        bool done = false;
        while (!done) {
            // ---- probe windows from run start `ip` ----
            int match = -1;
            {
                int k0 = 0;
                const int S0 = ip;
                bool found = false, aborted = false;
                while (true) {
                    int p[SNPW], pred[SNPW], maxgroup[SNPW];
                    uint32_t v[SNPW], h[SNPW];
                    uint16_t t[SNPW];
                    bool valid[SNPW], m[SNPW];
                    for (int l = 0; l < SNPW; l++) {
                        p[l] = S0 + (int)OFF[k0 + l];
                        int inc = (int)(OFF[k0 + l + 1] - OFF[k0 + l]);
                        valid[l] = p[l] + inc <= ip_limit;  // abort: next_ip > ip_limit
                        v[l] = (p[l] >= 0 && p[l] + 4 <= (int)n) ? rd32i(p[l]) : 0;
                        h[l] = snp_hash(v[l], shift);
                        t[l] = table[h[l]];
                    }
                    for (int l = 0; l < SNPW; l++) table[h[l]] = (uint16_t)l;
                    for (int l = 0; l < SNPW; l++) maxgroup[l] = table[h[l]];
                    for (int l = 0; l < SNPW; l++) pred[l] = -1;
                    {
                        uint64_t G = 0;
                        for (int l = 0; l < SNPW; l++) if (maxgroup[l] != l) G |= 1ULL << l;
                        while (G) {
                            int g = __builtin_ctzll(G);
                            int mg = maxgroup[g];
                            uint64_t members = 0;
                            for (int l = 0; l < SNPW; l++) if (maxgroup[l] == mg) members |= 1ULL << l;
                            for (int l = 0; l < SNPW; l++)
                                if (maxgroup[l] == mg) {
                                    uint64_t below = members & ((1ULL << l) - 1);
                                    if (below) pred[l] = 63 - __builtin_clzll(below);
                                }
                            G &= ~members;
                        }
                    }
                    int cand[SNPW];
                    for (int l = 0; l < SNPW; l++) {
                        cand[l] = pred[l] >= 0 ? p[pred[l]] : (int)t[l];
                        uint32_t cv = pred[l] >= 0 ? v[pred[l]] : rd32i((int)t[l]);
                        m[l] = valid[l] && cv == v[l];
                    }
                    int first_abort = SNPW, first_event = SNPW;
                    for (int l = 0; l < SNPW; l++) if (!valid[l]) { first_abort = l; break; }
                    for (int l = 0; l < SNPW; l++) if (m[l]) { first_event = l; break; }
                    bool have_match = first_event < first_abort && first_event < SNPW;
                    int commit_hi = have_match ? first_event
                                               : (first_abort < SNPW ? first_abort - 1 : SNPW - 1);
                    for (int l = 0; l < SNPW; l++) table[h[l]] = t[l];            // restore
                    for (int l = 0; l <= commit_hi; l++) table[h[l]] = (uint16_t)p[l];  // commit
                    if (have_match) {
                        ip = p[first_event];
                        match = cand[first_event];
                        found = true;
                    } else if (first_abort < SNPW) {
                        aborted = true;
                    }
                    if (found || aborted) break;
                    k0 += SNPW;
                }
                if (aborted) goto emit_remainder;
            }
            // ---- literal ----
            op = snp_emit_literal(op, s + next_emit, ip - next_emit);
            // ---- copy loop (uniform) ----
            {
                uint64_t input_bytes = 0;
                while (true) {
                    int base = ip;
                    int matched = 4 + snp_match_length(s + match + 4, s + ip + 4, s + n);
                    ip += matched;
                    op = snp_emit_copy(op, (size_t)(base - match), matched);
                    next_emit = ip;
                    if (ip >= ip_limit) goto emit_remainder;
                    input_bytes = snp_load64(s + ip - 1);
                    uint32_t prev_hash = snp_hash((uint32_t)input_bytes, shift);
                    table[prev_hash] = (uint16_t)(ip - 1);
                    uint32_t cur_hash = snp_hash((uint32_t)(input_bytes >> 8), shift);
                    int candidate = table[cur_hash];
                    uint32_t candidate_bytes = rd32i(candidate);
                    table[cur_hash] = (uint16_t)ip;
                    if ((uint32_t)(input_bytes >> 8) != candidate_bytes) break;
                    match = candidate;
                }
                ip++;  // next run starts at ip+1 (model: next_hash = ..>>16; ip++)
            }
            (void)done;
        }
    }
emit_remainder:
    if (next_emit < (int)n)
        op = snp_emit_literal(op, s + next_emit, (int)n - next_emit);
    return (int)(op - dst);
}
