// System libsnappy via dlopen (stable C ABI; no link-time dependency — the
// container's copy lives under /opt/conda/lib whose libstdc++ must NOT enter
// our link path).
// Pinning caveat: the reference bundles snappy
// 1.1.10 via snappy-java 1.1.10.4; this container has libsnappy 1.1.8. The
// snappy FORMAT is stable and decompress-equality holds across versions (the
// reference's own CompressorTest pins only that), but compressed-byte parity
// for Snappy sstables is pinned to 1.1.8 behaviour here.
#pragma once
#include <cstddef>
#include <dlfcn.h>
#include <stdexcept>

struct SnappyLib {
    int (*compress)(const char*, size_t, char*, size_t*);
    int (*uncompress)(const char*, size_t, char*, size_t*);
    size_t (*max_compressed_length)(size_t);
    int (*uncompressed_length)(const char*, size_t, size_t*);
};

inline const SnappyLib& snappy_lib() {
    static SnappyLib L = [] {
        SnappyLib l{};
        void* h = dlopen("libsnappy.so.1", RTLD_NOW | RTLD_LOCAL);
        if (!h) h = dlopen("/opt/conda/lib/libsnappy.so.1", RTLD_NOW | RTLD_LOCAL);
        if (!h) throw std::runtime_error("libsnappy.so.1 not found (Snappy sstables unsupported here)");
        l.compress = (int (*)(const char*, size_t, char*, size_t*))dlsym(h, "snappy_compress");
        l.uncompress = (int (*)(const char*, size_t, char*, size_t*))dlsym(h, "snappy_uncompress");
        l.max_compressed_length = (size_t (*)(size_t))dlsym(h, "snappy_max_compressed_length");
        l.uncompressed_length = (int (*)(const char*, size_t, size_t*))dlsym(h, "snappy_uncompressed_length");
        if (!l.compress || !l.uncompress || !l.max_compressed_length || !l.uncompressed_length)
            throw std::runtime_error("libsnappy C ABI symbols missing");
        return l;
    }();
    return L;
}

inline bool snappy_ref_compress(const char* in, size_t n, char* out, size_t* out_len) {
    return snappy_lib().compress(in, n, out, out_len) == 0;
}
inline bool snappy_ref_uncompressed_length(const char* in, size_t n, size_t* out) {
    return snappy_lib().uncompressed_length(in, n, out) == 0;
}
inline bool snappy_ref_uncompress(const char* in, size_t n, char* out) {
    size_t cap = 0;
    if (snappy_lib().uncompressed_length(in, n, &cap) != 0) return false;
    size_t got = cap;
    return snappy_lib().uncompress(in, n, out, &got) == 0 && got == cap;
}
inline size_t snappy_ref_max_compressed_length(size_t n) {
    return snappy_lib().max_compressed_length(n);
}
