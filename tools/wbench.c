// Write-pattern microbench: N threads pwrite disjoint ranges of ONE file
// (the Data.db drain pattern) on a given directory. Usage:
//   wbench <dir> <threads> <total_mb> [prealloc]
#define _GNU_SOURCE
#include <fcntl.h>
#include <pthread.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>
#include <unistd.h>

static char path[4096];
static size_t total, per;
static int prealloc;

static void* wr(void* a) {
    long t = (long)a;
    int fd = open(path, O_WRONLY);
    if (fd < 0) { perror("open"); return 0; }
    char* buf = malloc(16 << 20);
    memset(buf, 0x5A ^ t, 16 << 20);
    size_t off = t * per;
    for (size_t o = 0; o < per; o += 16 << 20) {
        size_t len = per - o < (16 << 20) ? per - o : (16 << 20);
        if (pwrite(fd, buf, len, off + o) != (ssize_t)len) perror("pwrite");
    }
    close(fd);
    free(buf);
    return 0;
}

int main(int argc, char** argv) {
    if (argc < 4) return 2;
    snprintf(path, sizeof path, "%s/wbench.bin", argv[1]);
    int nth = atoi(argv[2]);
    total = (size_t)atol(argv[3]) << 20;
    prealloc = argc > 4;
    per = total / nth;
    unlink(path);
    int fd = open(path, O_WRONLY | O_CREAT, 0644);
    if (prealloc && ftruncate(fd, total)) perror("ftruncate");
    close(fd);
    struct timespec a, b;
    clock_gettime(CLOCK_MONOTONIC, &a);
    pthread_t th[64];
    for (long t = 0; t < nth; t++) pthread_create(&th[t], 0, wr, (void*)t);
    for (int t = 0; t < nth; t++) pthread_join(th[t], 0);
    clock_gettime(CLOCK_MONOTONIC, &b);
    double s = (b.tv_sec - a.tv_sec) + (b.tv_nsec - a.tv_nsec) / 1e9;
    printf("%s th=%d %s: %.2f GB/s (%.0f ms for %zu MB)\n", argv[1], nth,
           prealloc ? "prealloc" : "grow", total / s / 1e9, s * 1e3, total >> 20);
    unlink(path);
    return 0;
}
