/*
 * q6_host.c — plain-C host driver over the cstripe C ABI, demonstrating the
 * north-star shape: "host code stays C ... and calls hand-written HIP kernels
 * through a thin C-ABI". This is the role the Postgres extension's C code
 * plays in the reference (ColumnarScanNext, columnar_customscan.c:1854-1894);
 * see INTEGRATION.md for the extension-side stub.
 *
 * Build:  gcc -O2 tools/q6_host.c -Lcitus_amd -lcstripe \
 *             -Wl,-rpath,'$ORIGIN/../citus_amd' -Iinclude -o tools/q6_host
 * Usage:  tools/q6_host <stripe-file> [steps]
 */
#define _POSIX_C_SOURCE 199309L
#include <stdio.h>
#include <stdlib.h>
#include <time.h>

#include "cstripe.h"

static double now_ms(void)
{
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec * 1e3 + ts.tv_nsec / 1e6;
}

int main(int argc, char **argv)
{
    if (argc < 2) {
        fprintf(stderr, "usage: %s <stripe-file> [steps]\n", argv[0]);
        return 2;
    }
    int steps = argc > 2 ? atoi(argv[2]) : 3;

    cstripe_reader *r = cstripe_open(argv[1]);
    if (!r) { fprintf(stderr, "open: %s\n", cstripe_errmsg()); return 1; }
    printf("rows=%llu cols=%u stripes=%u\n",
           (unsigned long long)cstripe_row_count(r),
           cstripe_column_count(r), cstripe_stripe_count(r));

    /* TPC-H Q6: shipdate in [1994-01-01, 1995-01-01), discount in [0.05,0.07],
     * quantity < 24; revenue = sum(extendedprice * discount) */
    cstripe_pred preds[5] = {
        {5, CSTRIPE_PRED_GE, 8766, 0}, {5, CSTRIPE_PRED_LT, 9131, 0},
        {3, CSTRIPE_PRED_GE, 5, 0},    {3, CSTRIPE_PRED_LE, 7, 0},
        {1, CSTRIPE_PRED_LT, 2400, 0},
    };
    cstripe_agg_spec aggs[2] = {
        {CSTRIPE_AGG_SUM_PROD_I64, 2, 3, -1, 0},
        {CSTRIPE_AGG_COUNT_STAR, -1, -1, -1, 0},
    };
    uint64_t proj = (1u << 2) | (1u << 3);   /* agg cols; pred cols auto-added */

    cstripe_scan *s = cstripe_scan_begin(r, proj, preds, 5);
    if (!s) { fprintf(stderr, "scan_begin: %s\n", cstripe_errmsg()); return 1; }
    printf("chunk groups pruned: %lld\n",
           (long long)cstripe_scan_chunk_groups_filtered(s));

    int rc = cstripe_gpu_stage(s, -1);
    if (rc != CSTRIPE_OK) { fprintf(stderr, "stage: %s\n", cstripe_errmsg()); return 1; }
    printf("staged %.3f GB (compressed) to HBM\n",
           cstripe_gpu_staged_bytes(s) / 1e9);

    cstripe_partial parts[2];
    for (int i = 0; i < steps; i++) {
        double t0 = now_ms();
        rc = cstripe_scan_agg(s, aggs, 2, parts);
        if (rc != CSTRIPE_OK) { fprintf(stderr, "scan_agg: %s\n", cstripe_errmsg()); return 1; }
        double dt = now_ms() - t0;
        printf("step %d: %.3f ms (decode %.3f + agg %.3f kernel ms), "
               "%.2f Grows/s\n", i, dt,
               cstripe_scan_last_decode_kernel_ms(s),
               cstripe_scan_last_agg_kernel_ms(s),
               cstripe_row_count(r) / dt / 1e6);
    }

    /* the combine step (single shard here, still through the surface) */
    cstripe_partial final[2];
    cagg_combine(aggs, 2, parts, 1, final);
    printf("Q6 revenue = %lld.%04lld (scale 4), count = %lld\n",
           (long long)(final[0].i128_lo / 10000),
           (long long)(final[0].i128_lo % 10000 < 0 ? -(final[0].i128_lo % 10000)
                                                    : final[0].i128_lo % 10000),
           (long long)final[1].count);

    cstripe_scan_end(s);
    cstripe_close(r);
    return 0;
}
