// System liblz4 1.9.3 prototypes (no lz4.h dev header in this image; the
// runtime library /usr/lib/x86_64-linux-gnu/liblz4.so.1.9.3 is present and is
// the SAME version the reference bundles via lz4-java 1.8.0 — BASELINE.md).
#pragma once
extern "C" {
int LZ4_compress_default(const char* src, char* dst, int srcSize, int dstCapacity);
int LZ4_decompress_safe(const char* src, char* dst, int compressedSize, int dstCapacity);
int LZ4_compressBound(int inputSize);
const char* LZ4_versionString(void);
}
