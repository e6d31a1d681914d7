// CPU pin: cassandra_amd/csrc/snappy_model.h (the snappy 1.1.8 compressor
// restatement the wave kernel must match) vs the SYSTEM libsnappy, byte for
// byte, across content patterns / sizes; plus round-trip through libsnappy's
// decompressor. No GPU.
#include "../../cassandra_amd/csrc/snappy_model.h"
#include "../../oracle/src/snappy_ref.h"

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

static uint64_t sm(uint64_t x) {
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

int main() {
    using namespace gpuc;
    std::vector<uint16_t> table(SNP_MAX_TABLE);
    int fails = 0, cases = 0;
    for (int mode = 0; mode < 7; mode++) {
        for (int trial = 0; trial < 60; trial++) {
            uint64_t seed = mode * 1000 + trial;
            uint32_t n = trial % 3 == 1 ? 1 + (uint32_t)(sm(seed) % 16384) : 16384;
            if (trial % 7 == 3) n = 1 + (uint32_t)(sm(seed ^ 77) % 64);  // tiny
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
            // reference: system libsnappy
            std::vector<uint8_t> ref(snappy_ref_max_compressed_length(n));
            size_t ref_len = ref.size();
            if (!snappy_ref_compress((const char*)src.data(), n, (char*)ref.data(), &ref_len)) {
                printf("libsnappy compress failed\n");
                return 2;
            }
            // model
            std::vector<uint8_t> got(snappy_ref_max_compressed_length(n) + 8);
            uint32_t ts = snp_table_size(n);
            int got_len = snp_compress(src.data(), n, got.data(), table.data(), ts);
            cases++;
            if ((size_t)got_len != ref_len || memcmp(got.data(), ref.data(), ref_len) != 0) {
                size_t d = 0;
                size_t lim = std::min((size_t)got_len, ref_len);
                while (d < lim && got[d] == ref[d]) d++;
                printf("MISMATCH mode=%d trial=%d n=%u: model=%d ref=%zu firstdiff=%zu "
                       "(model %02x ref %02x)\n",
                       mode, trial, n, got_len, ref_len, d,
                       d < (size_t)got_len ? got[d] : 0xEE, d < ref_len ? ref[d] : 0xEE);
                if (++fails > 4) return 1;
            }
        }
    }
    printf(fails ? "snappy model FAILED (%d/%d)\n" : "snappy model OK (%d cases)\n",
           fails ? fails : cases, cases);
    return fails ? 1 : 0;
}
