// CPU: wave-window decomposition (snappy_sim.h) vs the pinned scalar model
// (snappy_model.h) — byte-for-byte across the content sweep.
#include "snappy_sim.h"
#include <cstdio>
#include <vector>

static uint64_t sm(uint64_t x) {
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

int main() {
    using namespace gpuc;
    std::vector<uint16_t> ta(SNP_MAX_TABLE), tb(SNP_MAX_TABLE);
    int fails = 0, cases = 0;
    for (int mode = 0; mode < 7; mode++) {
        for (int trial = 0; trial < 60; trial++) {
            uint64_t seed = mode * 1000 + trial;
            uint32_t n = trial % 3 == 1 ? 1 + (uint32_t)(sm(seed) % 16384) : 16384;
            if (trial % 7 == 3) n = 1 + (uint32_t)(sm(seed ^ 77) % 64);
            std::vector<uint8_t> src(n);
            for (uint32_t i = 0; i < n; i++) {
                uint64_t r = sm(seed * 1315423911ULL + (i / 8));
                switch (mode) {
                    case 0: src[i] = (uint8_t)sm(seed + i); break;
                    case 1: src[i] = 0; break;
                    case 2: src[i] = (uint8_t)(r >> (8 * (i % 8))); break;
                    case 3: src[i] = (i % 3) ? (uint8_t)('a' + (i % 17)) : (uint8_t)sm(seed + i); break;
                    case 4: src[i] = (uint8_t)('a' + (sm(seed + i / 4) % 26)); break;
                    case 5: src[i] = (i % 512 == 0) ? (uint8_t)sm(seed + i) : 0x42; break;
                    default: src[i] = (uint8_t)((i / 100) & 0xFF); break;
                }
            }
            std::vector<uint8_t> a(4 + 2 * n + 64), b(4 + 2 * n + 64);
            uint32_t ts = snp_table_size(n);
            int la = snp_compress(src.data(), n, a.data(), ta.data(), ts);
            int lb = snp_sim_compress(src.data(), n, b.data(), tb.data(), ts);
            cases++;
            if (la != lb || memcmp(a.data(), b.data(), la) != 0) {
                size_t d = 0;
                while (d < (size_t)std::min(la, lb) && a[d] == b[d]) d++;
                printf("SIM MISMATCH mode=%d trial=%d n=%u model=%d sim=%d diff@%zu\n",
                       mode, trial, n, la, lb, d);
                if (++fails > 4) return 1;
            }
        }
    }
    printf(fails ? "snappy sim FAILED\n" : "snappy sim OK (%d cases)\n", cases);
    return fails ? 1 : 0;
}
