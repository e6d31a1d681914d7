// GPU test: wave-cooperative LZ4 kernels vs system liblz4 1.9.3 (compress
// bytes must be identical; decompress must round-trip). Build with hipcc.
#include <hip/hip_runtime.h>
#include "../../cassandra_amd/csrc/codec.h"
#include "../../cassandra_amd/csrc/gpu_structs.h"
#include "../../cassandra_amd/csrc/lz4_model.h"
// pull in kernels (single TU, same as the product build)
#include "../../cassandra_amd/csrc/kernels.hip"
#include "../../oracle/src/lz4_ref.h"
#include "lz4_sim.h"

#include <cstdio>
#include <cstring>
#include <vector>

using namespace gpuc;

static uint64_t sm(uint64_t x) { return splitmix64(x); }

int main() {
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev < 1) { printf("no gpu\n"); return 2; }
    uint32_t tab[256];
    crc32_make_table(tab);
    static uint32_t tab8[8 * 256];
    crc32_make_table8(tab8);  // decompress kernel uses the sliced set
    uint32_t* d_tab;
    hipMalloc(&d_tab, sizeof(tab8));
    hipMemcpy(d_tab, tab8, sizeof(tab8), hipMemcpyHostToDevice);
    // probe LDS order
    {
        unsigned int* d_p;
        hipMalloc(&d_p, 8);
        hipLaunchKernelGGL(k_probe_lds_order, dim3(1), dim3(64), 0, 0, d_p);
        unsigned int p[2];
        hipDeviceSynchronize();
        hipMemcpy(p, d_p, 8, hipMemcpyDeviceToHost);
        printf("lds write order probe: same-addr winner lane=%u reversed=%u (expect 63, 0)\n", p[0], p[1]);
        if (p[0] != 63 || p[1] != 0) { printf("LDS ORDER ASSUMPTION FAILED\n"); return 3; }
    }

    const int NCASE = 7, TRIALS = 40;
    int fails = 0, cases = 0;
    std::vector<uint8_t> src(CHUNK_LEN), ref(1 << 16);
    std::vector<uint16_t> table(LZ4M_HASHTABLESIZE_U16);
    for (int mode = 0; mode < NCASE; mode++) {
        for (int trial = 0; trial < TRIALS; trial++) {
            uint64_t seed = mode * 1000 + trial;
            uint32_t n = trial % 4 == 1 ? 1 + (uint32_t)(sm(seed) % CHUNK_LEN) : CHUNK_LEN;
            src.resize(n);
            for (uint32_t i = 0; i < n; i++) {
                uint64_t r = sm(seed * 1315423911ULL + (i / 8));
                switch (mode) {
                    case 0: src[i] = (uint8_t)sm(seed + i); break;                      // random
                    case 1: src[i] = 0; break;                                         // zeros
                    case 2: src[i] = (uint8_t)(r >> (8 * (i % 8))); break;             // repeat8
                    case 3: src[i] = (i % 3) ? 'a' + (i % 17) : (uint8_t)sm(seed + i); break;
                    case 4: src[i] = 'a' + (uint8_t)(sm(seed + i / 4) % 26); break;    // ascii
                    case 5: src[i] = (i % 512 == 0) ? (uint8_t)sm(seed + i) : 0x42; break;
                    default: src[i] = (uint8_t)((i / 100) & 0xFF); break;              // slow ramp
                }
            }
            // reference
            int rs = LZ4_compress_default((const char*)src.data(), (char*)ref.data(), (int)n,
                                          (int)ref.size());
            // gpu
            uint8_t *d_src, *d_slots;
            uint32_t *d_csize, *d_ccrc;
            hipMalloc(&d_src, n);
            hipMalloc(&d_slots, LZ4_SLOT);
            hipMalloc(&d_csize, 4);
            hipMalloc(&d_ccrc, 4);
            hipMemcpy(d_src, src.data(), n, hipMemcpyHostToDevice);
            hipLaunchKernelGGL(k_lz4_compress_wave_t<true>, dim3(1), dim3(64), 0, 0, d_src, (uint64_t)n,
                               d_slots, d_csize, d_ccrc, 1, d_tab);
            hipError_t e = hipDeviceSynchronize();
            if (e != hipSuccess) { printf("kernel error %s mode=%d trial=%d\n", hipGetErrorString(e), mode, trial); return 4; }
            uint32_t csz;
            hipMemcpy(&csz, d_csize, 4, hipMemcpyDeviceToHost);
            std::vector<uint8_t> gout(csz);
            hipMemcpy(gout.data(), d_slots, csz, hipMemcpyDeviceToHost);
            // compare (skip the 4-byte LE header)
            bool ok = (int)(csz - 4) == rs && memcmp(gout.data() + 4, ref.data(), rs) == 0;
            if (!ok) {
                int d = 0;
                int mn = std::min((int)csz - 4, rs);
                while (d < mn && gout[4 + d] == ref[d]) d++;
                printf("COMPRESS MISMATCH mode=%d trial=%d n=%u ref=%d gpu=%u first_diff=%d "
                       "(ref=%02x gpu=%02x)\n", mode, trial, n, rs, csz - 4, d,
                       d < rs ? ref[d] : 0, d < (int)csz - 4 ? gout[4 + d] : 0);
                // episode-level diff: rerun GPU with debug log + sim with episodes
                std::vector<uint32_t> seps;
                {
                    std::vector<uint8_t> tmp(LZ4_SLOT);
                    memset(table.data(), 0, table.size() * 2);
                    sim_compress(src.data(), (int)n, tmp.data(), table.data(), &seps);
                }
                uint32_t* d_dbg;
                hipMalloc(&d_dbg, 4 * 3 * 2001);
                hipMemset(d_dbg, 0, 4 * 3 * 2001);
                hipLaunchKernelGGL(k_lz4_compress_wave_t<true>, dim3(1), dim3(64), 0, 0, d_src, (uint64_t)n,
                                   d_slots, d_csize, d_ccrc, 1, d_tab, d_dbg);
                hipDeviceSynchronize();
                std::vector<uint32_t> geps(3 * 2001);
                hipMemcpy(geps.data(), d_dbg, 4 * 3 * 2001, hipMemcpyDeviceToHost);
                hipFree(d_dbg);
                uint32_t gn = geps[0], sn = (uint32_t)seps.size() / 3;
                printf("episodes: sim=%u gpu=%u\n", sn, gn);
                for (uint32_t e = 0; e < std::min({gn, sn, 2000u}); e++) {
                    uint32_t gip = geps[(e + 1) * 3], gma = geps[(e + 1) * 3 + 1], gli = geps[(e + 1) * 3 + 2];
                    uint32_t sip = seps[e * 3], sma = seps[e * 3 + 1], sli = seps[e * 3 + 2];
                    if (gip != sip || gma != sma || gli != sli) {
                        printf("episode %u: sim(ip=%u match=%u lit=%u) gpu(ip=%u match=%u lit=%u)\n",
                               e, sip, sma, sli, gip, gma, gli);
                        if (e > 0)
                            printf("  prev: sim(ip=%u match=%u lit=%u)\n", seps[(e-1)*3], seps[(e-1)*3+1], seps[(e-1)*3+2]);
                        break;
                    }
                }
                if (++fails > 2) return 1;
            }
            // decompress round trip via wave kernel
            {
                std::vector<uint8_t> frame = gout;  // header + block
                uint32_t crc = crc32_update_t(0, frame.data(), frame.size(), tab);
                for (int i = 3; i >= 0; i--) frame.push_back((uint8_t)(crc >> (8 * i)));
                uint8_t *d_comp, *d_out;
                unsigned long long* d_err;
                hipMalloc(&d_comp, frame.size());
                hipMalloc(&d_out, n);
                hipMalloc(&d_err, 8);
                hipMemset(d_err, 0, 8);
                hipMemcpy(d_comp, frame.data(), frame.size(), hipMemcpyHostToDevice);
                ChunkDesc cd{d_comp, d_out, (uint32_t)frame.size() - 4, n};
                ChunkDesc* d_cd;
                hipMalloc(&d_cd, sizeof(cd));
                hipMemcpy(d_cd, &cd, sizeof(cd), hipMemcpyHostToDevice);
                hipLaunchKernelGGL(k_lz4_decompress_wave, dim3(lz4_decomp_grid(1, 1)), dim3(64), 0, 0, d_cd, 1, 1, d_err, d_tab);
                hipDeviceSynchronize();
                unsigned long long err;
                hipMemcpy(&err, d_err, 8, hipMemcpyDeviceToHost);
                std::vector<uint8_t> rt(n);
                hipMemcpy(rt.data(), d_out, n, hipMemcpyDeviceToHost);
                if (err || memcmp(rt.data(), src.data(), n) != 0) {
                    printf("DECOMPRESS FAIL mode=%d trial=%d err=%llu\n", mode, trial, err);
                    if (++fails > 5) return 1;
                }
                hipFree(d_comp); hipFree(d_out); hipFree(d_err); hipFree(d_cd);
            }
            hipFree(d_src); hipFree(d_slots); hipFree(d_csize); hipFree(d_ccrc);
            cases++;
        }
    }
    printf(fails ? "FAILED %d of %d\n" : "lz4 wave OK (%d cases)\n", fails ? fails : cases, cases);
    if (fails) return 1;

    // ---- A/B throughput: LDS-staged vs global-chunk, content sweep ----
    for (int content = 0; content < 4; content++) {
        const uint32_t NC = 16384;  // 256 MiB of chunks
        std::vector<uint8_t> big((uint64_t)NC * CHUNK_LEN);
        for (uint64_t i = 0; i < big.size(); i += 8) {
            uint64_t r = sm(i);
            uint64_t w;
            switch (content) {
                case 0: w = (r % 100 < 55 && i) ? *(uint64_t*)&big[i - 8] : sm(r); break;
                case 1: w = sm(r); break;        // incompressible
                case 2: w = 0; break;            // all zeros (1 long match)
                default: w = (r % 100 < 97 && i) ? *(uint64_t*)&big[i - 8] : sm(r); break;
            }
            memcpy(&big[i], &w, 8);
        }
        printf("content=%d (%s)\n", content,
               content == 0 ? "vrep55" : content == 1 ? "random" : content == 2 ? "zeros" : "vrep97");
        uint8_t *d_big, *d_slots2;
        uint32_t *d_cs, *d_crc2;
        hipMalloc(&d_big, big.size());
        hipMalloc(&d_slots2, (uint64_t)NC * LZ4_SLOT);
        hipMalloc(&d_cs, NC * 4);
        hipMalloc(&d_crc2, NC * 4);
        hipMemcpy(d_big, big.data(), big.size(), hipMemcpyHostToDevice);
        for (int variant = 0; variant < 2; variant++) {
            hipEvent_t a, b;
            hipEventCreate(&a); hipEventCreate(&b);
            // warmup
            if (variant == 0)
                hipLaunchKernelGGL(k_lz4_compress_wave_t<true>, dim3(NC), dim3(64), 0, 0, d_big, big.size(), d_slots2, d_cs, d_crc2, NC, d_tab);
            else
                hipLaunchKernelGGL(k_lz4_compress_wave_t<false>, dim3(NC), dim3(64), 0, 0, d_big, big.size(), d_slots2, d_cs, d_crc2, NC, d_tab);
            hipDeviceSynchronize();
            hipEventRecord(a);
            for (int it = 0; it < 3; it++) {
                if (variant == 0)
                    hipLaunchKernelGGL(k_lz4_compress_wave_t<true>, dim3(NC), dim3(64), 0, 0, d_big, big.size(), d_slots2, d_cs, d_crc2, NC, d_tab);
                else
                    hipLaunchKernelGGL(k_lz4_compress_wave_t<false>, dim3(NC), dim3(64), 0, 0, d_big, big.size(), d_slots2, d_cs, d_crc2, NC, d_tab);
            }
            hipEventRecord(b);
            hipDeviceSynchronize();
            float ms;
            hipEventElapsedTime(&ms, a, b);
            printf("compress %s: %.1f ms for 3x256MiB = %.2f GB/s in\n",
                   variant == 0 ? "LDS-staged " : "global-src", ms, 3.0 * big.size() / (ms / 1e3) / 1e9);
        }
        hipFree(d_big); hipFree(d_slots2); hipFree(d_cs); hipFree(d_crc2);
    }
    return 0;
}
