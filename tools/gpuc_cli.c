/* Pure-C consumer of the libcassandra_gpucompact C ABI — demonstrates the
 * drop-in boundary (include/gpucompact.h) without C++ or Python. The Java
 * side would bind the same symbols via Panama FFI.
 *
 *   gpuc_cli compact <out_base> <in_base> [<in_base> ...]
 *   gpuc_cli verify  <base>
 *   gpuc_cli scrub   <out_base> <in_base>
 */
#include <stdio.h>
#include <stdint.h>
#include <string.h>
#include "../include/gpucompact.h"

int main(int argc, char** argv) {
    char err[256] = {0};
    if (argc < 3) {
        fprintf(stderr, "usage: %s compact|verify|scrub ...\n", argv[0]);
        return 2;
    }
    printf("%s | devices: %d\n", gpuc_version(), gpuc_device_count());
    if (!strcmp(argv[1], "verify")) {
        int rc = gpuc_verify(argv[2], 0, err, sizeof err);
        printf("verify rc=%d %s\n", rc, err);
        return rc;
    }
    if (!strcmp(argv[1], "scrub")) {
        uint64_t kept = 0, dropped = 0;
        int rc = gpuc_scrub(argv[3], argv[2], 0, &kept, &dropped, err, sizeof err);
        printf("scrub rc=%d kept=%llu dropped=%llu %s\n", rc,
               (unsigned long long)kept, (unsigned long long)dropped, err);
        return rc;
    }
    if (!strcmp(argv[1], "compact")) {
        const char* ins[64];
        int k = argc - 3;
        if (k < 1 || k > 64) { fprintf(stderr, "1..64 inputs\n"); return 2; }
        for (int i = 0; i < k; i++) ins[i] = argv[3 + i];
        gpuc_job job;
        memset(&job, 0, sizeof job);
        job.input_bases = ins;
        job.n_inputs = (uint32_t)k;
        job.output_base = argv[2];
        job.gc_before = INT64_MIN;
        job.never_purge = 1;
        gpuc_result res;
        memset(&res, 0, sizeof res);
        int rc = gpuc_compact(&job, &res);
        printf("compact rc=%d in=%llu B out_parts=%llu rows=%llu %s\n", rc,
               (unsigned long long)res.input_uncompressed_bytes,
               (unsigned long long)res.partitions_out,
               (unsigned long long)res.rows_out, res.error);
        return rc;
    }
    fprintf(stderr, "unknown command %s\n", argv[1]);
    return 2;
}
