// Test: fuzz the product's LZ4 model (cassandra_amd/csrc/lz4_model.h) against
// the system liblz4 1.9.3 (== the reference's bundled codec version).
#include "../../cassandra_amd/csrc/lz4_model.h"
#include "../../oracle/src/lz4_ref.h"
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <cstring>

static uint64_t sm64(uint64_t x){x+=0x9E3779B97f4A7C15ULL;x=(x^(x>>30))*0xBF58476D1CE4E5B9ULL;x=(x^(x>>27))*0x94D049BB133111EBULL;return x^(x>>31);}

int main() {
    std::vector<uint8_t> src, a, b;
    std::vector<uint16_t> table(LZ4M_HASHTABLESIZE_U16);
    int cases = 0;
    for (int mode = 0; mode < 6; mode++) {
        for (int trial = 0; trial < 200; trial++) {
            uint64_t seed = mode * 1000 + trial;
            int n;
            switch (trial % 5) {
                case 0: n = 16384; break;
                case 1: n = 1 + (int)(sm64(seed) % 16384); break;
                case 2: n = 12 + (int)(sm64(seed) % 30); break;  // around minLength boundary
                case 3: n = 1 + (int)(sm64(seed) % 13); break;
                default: n = 16384; break;
            }
            src.resize(n);
            // content modes: random / zero / repeat8 / mixed / ascii / sparse-change
            for (int i = 0; i < n; i++) {
                uint64_t r = sm64(seed * 1315423911ULL + (uint64_t)(i / 8));
                switch (mode) {
                    case 0: src[i] = (uint8_t)(sm64(seed + i) & 0xFF); break;
                    case 1: src[i] = 0; break;
                    case 2: src[i] = (uint8_t)(r >> (8 * (i % 8))); break;
                    case 3: src[i] = (i % 3) ? 'a' + (i % 17) : (uint8_t)sm64(seed + i); break;
                    case 4: src[i] = 'a' + (uint8_t)(sm64(seed + i / 4) % 26); break;
                    default: src[i] = (uint8_t)((i % 512 == 0) ? sm64(seed + i) : 0x42); break;
                }
            }
            int bound = LZ4_compressBound(n);
            a.assign(bound, 0xAA);
            b.assign(bound, 0xBB);
            int ra = LZ4_compress_default((const char*)src.data(), (char*)a.data(), n, bound);
            memset(table.data(), 0, table.size() * 2);
            int rb = lz4m_compress(src.data(), n, b.data(), table.data());
            if (ra != rb || memcmp(a.data(), b.data(), ra) != 0) {
                int d = 0;
                while (d < ra && d < rb && a[d] == b[d]) d++;
                printf("MISMATCH mode=%d trial=%d n=%d ref=%d model=%d first_diff=%d\n",
                       mode, trial, n, ra, rb, d);
                return 1;
            }
            cases++;
        }
    }
    printf("lz4 model fuzz OK (%d cases, liblz4 %s)\n", cases, LZ4_versionString());
    return 0;
}
