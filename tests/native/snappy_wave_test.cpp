// GPU test: wave-cooperative snappy kernels vs the CPU-pinned model
// (snappy_model.h == libsnappy 1.1.8 bytes): compress must be byte-identical;
// decompress must round-trip model-compressed chunks. Plus an A/B content
// throughput sweep. Build with hipcc; needs an MI355X.
#include <hip/hip_runtime.h>
#include "../../cassandra_amd/csrc/codec.h"
#include "../../cassandra_amd/csrc/gpu_structs.h"
#include "../../cassandra_amd/csrc/kernels.hip"
#include "snappy_sim.h"

#include <cstdio>
#include <cstring>
#include <vector>

using namespace gpuc;

#define HC(x) do { hipError_t e_ = (x); if (e_ != hipSuccess) { \
    printf("HIP error %s at %d\n", hipGetErrorString(e_), __LINE__); return 9; } } while (0)

static uint64_t sm(uint64_t x) {
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

static void fill(std::vector<uint8_t>& src, int mode, uint64_t seed) {
    for (size_t i = 0; i < src.size(); i++) {
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
}

int main() {
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev < 1) { printf("no gpu\n"); return 2; }
    const auto& OFFh = snp_off().off;
    uint32_t* d_off;
    HC(hipMalloc(&d_off, OFFh.size() * 4));
    HC(hipMemcpy(d_off, OFFh.data(), OFFh.size() * 4, hipMemcpyHostToDevice));
    std::vector<uint16_t> table(SNP_MAX_TABLE);

    // ---- correctness: batched compress vs model; decompress round-trip ----
    int fails = 0, cases = 0;
    for (int mode = 0; mode < 7; mode++) {
        const int NT = 40;
        std::vector<std::vector<uint8_t>> srcs(NT);
        std::vector<std::vector<uint8_t>> refs(NT);
        uint64_t total_in = 0, total_slot = 0;
        for (int t = 0; t < NT; t++) {
            uint64_t seed = mode * 1000 + t;
            uint32_t n = t % 3 == 1 ? 1 + (uint32_t)(sm(seed) % 16384) : 16384;
            if (t % 7 == 3) n = 1 + (uint32_t)(sm(seed ^ 77) % 64);
            srcs[t].resize(n);
            fill(srcs[t], mode, seed);
            refs[t].resize(snp_table_size(n) * 0 + 4 + 2 * n + 64);
            int rl = snp_compress(srcs[t].data(), n, refs[t].data(), table.data(), snp_table_size(n));
            refs[t].resize(rl);
            total_in += n;
            total_slot += 40960;
        }
        uint8_t *d_src, *d_dst, *d_rt;
        uint32_t* d_csize;
        unsigned long long* d_err;
        HC(hipMalloc(&d_src, total_in + 64));
        HC(hipMalloc(&d_dst, total_slot + 64));
        HC(hipMalloc(&d_rt, total_in + 64));
        HC(hipMalloc(&d_csize, NT * 4));
        HC(hipMalloc(&d_err, 8));
        HC(hipMemset(d_err, 0, 8));
        std::vector<SnpChunk> cc(NT), dc(NT);
        uint64_t io = 0, oo = 0;
        for (int t = 0; t < NT; t++) {
            HC(hipMemcpy(d_src + io, srcs[t].data(), srcs[t].size(), hipMemcpyHostToDevice));
            cc[t] = {d_src + io, d_dst + oo, (uint32_t)srcs[t].size()};
            io += srcs[t].size();
            oo += 40960;
        }
        SnpChunk* d_cc;
        HC(hipMalloc(&d_cc, NT * sizeof(SnpChunk)));
        HC(hipMemcpy(d_cc, cc.data(), NT * sizeof(SnpChunk), hipMemcpyHostToDevice));
        hipLaunchKernelGGL(k_snappy_compress_wave, dim3(NT), dim3(WAVE), 0, 0, d_cc, NT,
                           d_csize, d_off, (uint32_t)OFFh.size());
        HC(hipDeviceSynchronize());
        std::vector<uint32_t> csize(NT);
        HC(hipMemcpy(csize.data(), d_csize, NT * 4, hipMemcpyDeviceToHost));
        std::vector<uint8_t> got(40960);
        io = 0;
        for (int t = 0; t < NT; t++) {
            HC(hipMemcpy(got.data(), d_dst + (uint64_t)t * 40960, csize[t], hipMemcpyDeviceToHost));
            cases++;
            if (csize[t] != refs[t].size() || memcmp(got.data(), refs[t].data(), refs[t].size())) {
                size_t d = 0, lim = std::min((size_t)csize[t], refs[t].size());
                while (d < lim && got[d] == refs[t][d]) d++;
                printf("COMPRESS MISMATCH mode=%d t=%d n=%zu gpu=%u ref=%zu diff@%zu\n", mode, t,
                       srcs[t].size(), csize[t], refs[t].size(), d);
                if (++fails > 3) return 1;
            }
        }
        // decompress round trip of the model-compressed refs
        io = 0;
        oo = 0;
        for (int t = 0; t < NT; t++) {
            HC(hipMemcpy(d_dst + (uint64_t)t * 40960, refs[t].data(), refs[t].size(),
                         hipMemcpyHostToDevice));
            dc[t] = {d_dst + (uint64_t)t * 40960, d_rt + io, (uint32_t)refs[t].size()};
            io += srcs[t].size();
        }
        HC(hipMemcpy(d_cc, dc.data(), NT * sizeof(SnpChunk), hipMemcpyHostToDevice));
        hipLaunchKernelGGL(k_snappy_decompress_wave, dim3(NT), dim3(WAVE), 0, 0, d_cc, NT, d_err,
                           (uint8_t*)nullptr);
        HC(hipDeviceSynchronize());
        unsigned long long err = 0;
        HC(hipMemcpy(&err, d_err, 8, hipMemcpyDeviceToHost));
        if (err) { printf("DECOMPRESS ERROR code %llu mode=%d\n", err, mode); return 1; }
        io = 0;
        std::vector<uint8_t> rt(16384);
        for (int t = 0; t < NT; t++) {
            rt.resize(srcs[t].size());
            HC(hipMemcpy(rt.data(), d_rt + io, srcs[t].size(), hipMemcpyDeviceToHost));
            io += srcs[t].size();
            if (memcmp(rt.data(), srcs[t].data(), srcs[t].size())) {
                printf("ROUNDTRIP MISMATCH mode=%d t=%d\n", mode, t);
                if (++fails > 3) return 1;
            }
        }
        HC(hipFree(d_src)); HC(hipFree(d_dst)); HC(hipFree(d_rt));
        HC(hipFree(d_csize)); HC(hipFree(d_err)); HC(hipFree(d_cc));
    }
    printf(fails ? "snappy wave FAILED (%d)\n" : "snappy wave OK (%d cases + roundtrips)\n",
           fails ? fails : cases);
    if (fails) return 1;

    // ---- throughput sweep: 3x256 MiB per content ----
    for (int content = 0; content < 3; content++) {
        const uint64_t TOT = 3ull * 256 * 1024 * 1024;
        const uint32_t NC = (uint32_t)(TOT / 16384);
        std::vector<uint8_t> h(16384 * 64);
        fill(h, content == 0 ? 2 : content == 1 ? 0 : 4, 99 + content);
        uint8_t *d_src, *d_dst;
        uint32_t* d_csize;
        HC(hipMalloc(&d_src, h.size()));
        HC(hipMalloc(&d_dst, (uint64_t)NC * 40960));
        HC(hipMalloc(&d_csize, NC * 4));
        HC(hipMemcpy(d_src, h.data(), h.size(), hipMemcpyHostToDevice));
        std::vector<SnpChunk> cc(NC);
        for (uint32_t c = 0; c < NC; c++)
            cc[c] = {d_src + (uint64_t)(c % 64) * 16384, d_dst + (uint64_t)c * 40960, 16384};
        SnpChunk* d_cc;
        HC(hipMalloc(&d_cc, NC * sizeof(SnpChunk)));
        HC(hipMemcpy(d_cc, cc.data(), NC * sizeof(SnpChunk), hipMemcpyHostToDevice));
        hipEvent_t e0, e1;
        HC(hipEventCreate(&e0)); HC(hipEventCreate(&e1));
        hipLaunchKernelGGL(k_snappy_compress_wave, dim3(256), dim3(WAVE), 0, 0, d_cc, 256,
                           d_csize, d_off, (uint32_t)OFFh.size());  // warm
        HC(hipDeviceSynchronize());
        HC(hipEventRecord(e0, 0));
        hipLaunchKernelGGL(k_snappy_compress_wave, dim3(NC), dim3(WAVE), 0, 0, d_cc, NC, d_csize,
                           d_off, (uint32_t)OFFh.size());
        HC(hipEventRecord(e1, 0));
        HC(hipDeviceSynchronize());
        float ms = 0;
        HC(hipEventElapsedTime(&ms, e0, e1));
        printf("snappy compress %s: %.1f ms for 3x256MiB = %.2f GB/s in\n",
               content == 0 ? "vrep-ish" : content == 1 ? "zeros" : "ascii", ms,
               TOT / (ms * 1e6));
        HC(hipFree(d_src)); HC(hipFree(d_dst)); HC(hipFree(d_csize)); HC(hipFree(d_cc));
    }
    return 0;
}
