/*
 * bench_gen.cpp — seeded synthetic lineitem-shaped data generator
 * (bench/test infrastructure; value distributions per SURVEY.md §8d:
 * xorshift64 seed 42; quantity fixed-point [100,5000]; price [900,105000];
 * discount [0,10]; tax [0,8]; shipdate uniform over 7 years of days
 * (1992-01-01=8035 .. 1998-12-31=10591); returnflag A/N/R 25/25/50;
 * linestatus O/F 50/50). Physical schema (fixed-point int64 scale 2 for
 * decimal(15,2) measures; i8 categorical codes for char(1) flags):
 *   0 l_orderkey i64, 1 l_quantity i64/s2, 2 l_extendedprice i64/s2,
 *   3 l_discount i64/s2, 4 l_tax i64/s2, 5 l_shipdate i64 (days),
 *   6 l_returnflag i8 (0=A,1=N,2=R), 7 l_linestatus i8 (0=O,1=F)
 */
#include "../../include/cstripe.h"

#include <cstdint>
#include <cstring>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

extern "C" int csbench_gen_lineitem_mm(const char *path, uint64_t n_rows, uint64_t seed,
                                       int compression, int level, int seg_kb,
                                       uint64_t stripe_rows, uint32_t chunk_rows,
                                       int min_match);
extern "C" int csbench_gen_lineitem2(const char *path, uint64_t n_rows, uint64_t seed,
                                     int compression, int level, int seg_kb,
                                     uint64_t stripe_rows, uint32_t chunk_rows,
                                     int min_match, int canonical);

static inline uint64_t xs64(uint64_t &x)
{
    x ^= x << 13;
    x ^= x >> 7;
    x ^= x << 17;
    return x;
}

extern "C" int csbench_gen_lineitem(const char *path, uint64_t n_rows, uint64_t seed,
                                    int compression, int level, int seg_kb,
                                    uint64_t stripe_rows, uint32_t chunk_rows)
{
    return csbench_gen_lineitem_mm(path, n_rows, seed, compression, level, seg_kb,
                                   stripe_rows, chunk_rows, 0);
}

extern "C" int csbench_gen_lineitem_mm(const char *path, uint64_t n_rows, uint64_t seed,
                                       int compression, int level, int seg_kb,
                                       uint64_t stripe_rows, uint32_t chunk_rows,
                                       int min_match)
{
    return csbench_gen_lineitem2(path, n_rows, seed, compression, level, seg_kb,
                                 stripe_rows, chunk_rows, min_match, 1);
}

extern "C" int csbench_gen_lineitem2(const char *path, uint64_t n_rows, uint64_t seed,
                                     int compression, int level, int seg_kb,
                                     uint64_t stripe_rows, uint32_t chunk_rows,
                                     int min_match, int canonical)
{
    cstripe_coldef cols[8] = {};
    const char *names[8] = {"l_orderkey", "l_quantity", "l_extendedprice", "l_discount",
                            "l_tax", "l_shipdate", "l_returnflag", "l_linestatus"};
    for (int i = 0; i < 8; i++) {
        strncpy(cols[i].name, names[i], 31);
        cols[i].type = (i < 6) ? CSTRIPE_I64 : CSTRIPE_I8;
        cols[i].scale = (i >= 1 && i <= 4) ? 2 : 0;
    }
    cstripe_options opts;
    cstripe_default_options(&opts);
    if (stripe_rows) opts.stripe_row_limit = stripe_rows;
    if (chunk_rows) opts.chunk_group_row_limit = chunk_rows;
    opts.compression = (uint8_t)compression;
    opts.compression_level = (int8_t)level;
    if (seg_kb > 0) { opts.lz4_seg_target_kb = (uint16_t)seg_kb; opts.lz4_seg_target_bytes = 0; }
    else if (seg_kb < 0) { opts.lz4_seg_target_kb = 0; opts.lz4_seg_target_bytes = (uint32_t)(-seg_kb); }
    if (min_match >= 4) opts.lz4_min_match = (uint8_t)min_match;
    opts.canonical = canonical ? 1 : 0;

    cstripe_writer *w = cstripe_write_begin(path, cols, 8, &opts);
    if (!w) return CSTRIPE_ERR;

    const uint64_t seed0 = seed ? seed : 42;
    const uint64_t BATCH = 1u << 20;
    std::vector<int64_t> c0(BATCH), c1(BATCH), c2(BATCH), c3(BATCH), c4(BATCH), c5(BATCH);
    std::vector<int8_t> c6(BATCH), c7(BATCH);
    uint64_t done = 0, orderkey = 1;
    while (done < n_rows) {
        uint64_t n = n_rows - done < BATCH ? n_rows - done : BATCH;
        /* one independent xorshift64 stream per (column, batch) — columns
         * generate in parallel, deterministically for a given seed */
        const uint64_t b = done / BATCH;
        #pragma omp parallel for schedule(static)
        for (int col = 0; col < 7; col++) {
            uint64_t x = seed0 * 0x9E3779B97F4A7C15ull + b * 0xBF58476D1CE4E5B9ull
                         + (uint64_t)(col + 1) * 0x94D049BB133111EBull;
            xs64(x); xs64(x);                       /* warm the stream */
            switch (col) {
                case 0: for (uint64_t i = 0; i < n; i++) c1[i] = 100 + (int64_t)(xs64(x) % 4901); break;
                case 1: for (uint64_t i = 0; i < n; i++) c2[i] = 900 + (int64_t)(xs64(x) % 104101); break;
                case 2: for (uint64_t i = 0; i < n; i++) c3[i] = (int64_t)(xs64(x) % 11); break;
                case 3: for (uint64_t i = 0; i < n; i++) c4[i] = (int64_t)(xs64(x) % 9); break;
                case 4: for (uint64_t i = 0; i < n; i++) c5[i] = 8035 + (int64_t)(xs64(x) % 2557); break;
                case 5: for (uint64_t i = 0; i < n; i++) {
                            uint64_t rf = xs64(x) % 4;
                            c6[i] = (int8_t)(rf == 0 ? 0 : (rf == 1 ? 1 : 2));
                        }
                        break;
                case 6: for (uint64_t i = 0; i < n; i++) c7[i] = (int8_t)(xs64(x) % 2); break;
            }
        }
        for (uint64_t i = 0; i < n; i++) c0[i] = (int64_t)(orderkey + i);
        orderkey += n;
        const void *vals[8] = {c0.data(), c1.data(), c2.data(), c3.data(),
                               c4.data(), c5.data(), c6.data(), c7.data()};
        int rc = cstripe_write_rows(w, n, vals, nullptr);
        if (rc != CSTRIPE_OK) { cstripe_write_abort(w); return rc; }
        done += n;
    }
    return cstripe_write_end(w);
}
