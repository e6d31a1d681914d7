// CPU fuzz of the wave-algorithm simulator vs liblz4 (see lz4_sim.h).
#include "lz4_sim.h"
#include "../../oracle/src/lz4_ref.h"
#include <cstdio>
#include <vector>
#include <algorithm>

static uint64_t sm(uint64_t x){x+=0x9E3779B97f4A7C15ULL;x=(x^(x>>30))*0xBF58476D1CE4E5B9ULL;x=(x^(x>>27))*0x94D049BB133111EBULL;return x^(x>>31);}

void fill_case(std::vector<uint8_t>& src, int mode, int trial) {
    uint64_t seed = mode * 1000 + trial;
    int n = trial % 4 == 1 ? 1 + (int)(sm(seed) % 16384) : 16384;
    src.resize(n);
    for (int i = 0; i < n; i++) {
        uint64_t r = sm(seed * 1315423911ULL + (uint64_t)(i / 8));
        switch (mode) {
            case 0: src[i] = (uint8_t)sm(seed + i); break;
            case 1: src[i] = 0; break;
            case 2: src[i] = (uint8_t)(r >> (8 * (i % 8))); break;
            case 3: src[i] = (i % 3) ? 'a' + (i % 17) : (uint8_t)sm(seed + i); break;
            case 4: src[i] = 'a' + (uint8_t)(sm(seed + i / 4) % 26); break;
            case 5: src[i] = (i % 512 == 0) ? (uint8_t)sm(seed + i) : 0x42; break;
            case 6: src[i] = (uint8_t)((i / 100) & 0xFF); break;
            default: src[i] = (uint8_t)(sm(seed + i) % 4); break;
        }
    }
}

int main() {
    std::vector<uint8_t> src, a, b;
    std::vector<uint16_t> table(LZ4M_HASHTABLESIZE_U16);
    int fails = 0, cases = 0;
    for (int mode = 0; mode < 8; mode++) {
        for (int trial = 0; trial < 100; trial++) {
            fill_case(src, mode, trial);
            int n = (int)src.size();
            int bound = LZ4_compressBound(n);
            a.assign(bound, 0);
            b.assign(bound, 0);
            int ra = LZ4_compress_default((const char*)src.data(), (char*)a.data(), n, bound);
            memset(table.data(), 0, table.size() * 2);
            int rb = sim_compress(src.data(), n, b.data(), table.data());
            if (ra != rb || memcmp(a.data(), b.data(), ra)) {
                printf("SIM MISMATCH mode=%d trial=%d\n", mode, trial);
                if (++fails > 3) return 1;
            }
            cases++;
        }
    }
    printf(fails ? "FAILED %d\n" : "sim OK (%d cases)\n", fails ? fails : cases);
    return fails ? 1 : 0;
}
