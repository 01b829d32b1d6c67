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
        /* flags are REAL char(1) columns: short-varlena slots exactly as the
         * reference stores them (round-1 VERDICT #6 — no pre-coded bytes) */
        cols[i].type = (i < 6) ? CSTRIPE_I64 : CSTRIPE_TEXT;
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
    std::vector<uint32_t> c6(BATCH), c7(BATCH);
    /* 1-char short varlena slot: hdr (2<<1)|1 = 0x05, payload, zero pad */
    auto slot1 = [](char ch) { return 0x05u | ((uint32_t)(uint8_t)ch << 8); };
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
                            c6[i] = slot1(rf == 0 ? 'A' : (rf == 1 ? 'N' : 'R'));
                        }
                        break;
                case 6: for (uint64_t i = 0; i < n; i++)
                            c7[i] = slot1((xs64(x) % 2) == 0 ? 'O' : 'F');
                        break;
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

/* ================= round 2: parallel sharded gen + expected values =======
 * csbench_gen_lineitem_shards: writes n_shards files dir/shardNN.cs in
 * parallel (one writer thread per shard; seeds base_seed + i), the config-3
 * shard-directory shape. csbench_expected_q6/_q1: recompute the bench
 * queries' exact answers straight from the generator streams (int128 on
 * host, no file, parallel over batches) — the in-run parity pin for the
 * 1B-row headline (round-1 VERDICT #2). The stream constants here MUST stay
 * byte-identical to csbench_gen_lineitem2's.
 * ======================================================================= */
#include <thread>
#include <string>
#include <atomic>

extern "C" int csbench_gen_lineitem_shards(const char *dir, uint64_t total_rows,
                                           uint32_t n_shards, uint64_t base_seed,
                                           int compression, int level, int seg_kb,
                                           uint64_t stripe_rows, uint32_t chunk_rows,
                                           int min_match, int canonical, int threads)
{
    if (n_shards == 0) return CSTRIPE_ERR_ARG;
    if (threads <= 0) {
        threads = (int)std::thread::hardware_concurrency();
        if (threads <= 0) threads = 8;
    }
    std::atomic<uint32_t> next{0};
    std::atomic<int> rc{CSTRIPE_OK};
    const uint64_t per = total_rows / n_shards;
    auto worker = [&]() {
#ifdef _OPENMP
        /* shard workers ARE the parallelism: without this every worker
         * spawns its own full-width OMP team inside flush_stripe's
         * parallel-for -> thousands of threads, ~10x slower generation */
        omp_set_num_threads(1);
#endif
        for (;;) {
            uint32_t i = next.fetch_add(1);
            if (i >= n_shards || rc.load() != CSTRIPE_OK) return;
            char path[4096];
            snprintf(path, sizeof(path), "%s/shard%02u.cs", dir, i);
            uint64_t rows = per + (i == 0 ? total_rows % n_shards : 0);
            int r = csbench_gen_lineitem2(path, rows, base_seed + i, compression,
                                          level, seg_kb, stripe_rows, chunk_rows,
                                          min_match, canonical);
            if (r != CSTRIPE_OK) rc.store(r);
        }
    };
    int nw = (int)n_shards < threads ? (int)n_shards : threads;
    std::vector<std::thread> ts;
    for (int i = 0; i < nw; i++) ts.emplace_back(worker);
    for (auto &t : ts) t.join();
    return rc.load();
}

namespace {

struct ExpQ6 {
    __int128 revenue = 0;
    int64_t count = 0;
};

struct ExpQ1 {
    __int128 sum_qty[6] = {};
    __int128 sum_price[6] = {};
    __int128 sum_disc_price[6] = {};
    __int128 sum_charge[6] = {};
    int64_t count[6] = {};
};

/* regenerate one batch's Q-relevant columns exactly as the generator does */
void gen_batch(uint64_t seed0, uint64_t b, uint64_t n,
               std::vector<int64_t> &qty, std::vector<int64_t> &price,
               std::vector<int64_t> &disc, std::vector<int64_t> &tax,
               std::vector<int64_t> &ship, std::vector<int8_t> &rf,
               std::vector<int8_t> &ls)
{
    for (int col = 0; col < 7; col++) {
        uint64_t x = seed0 * 0x9E3779B97F4A7C15ull + b * 0xBF58476D1CE4E5B9ull
                     + (uint64_t)(col + 1) * 0x94D049BB133111EBull;
        xs64(x); xs64(x);
        switch (col) {
            case 0: for (uint64_t i = 0; i < n; i++) qty[i] = 100 + (int64_t)(xs64(x) % 4901); break;
            case 1: for (uint64_t i = 0; i < n; i++) price[i] = 900 + (int64_t)(xs64(x) % 104101); break;
            case 2: for (uint64_t i = 0; i < n; i++) disc[i] = (int64_t)(xs64(x) % 11); break;
            case 3: for (uint64_t i = 0; i < n; i++) tax[i] = (int64_t)(xs64(x) % 9); break;
            case 4: for (uint64_t i = 0; i < n; i++) ship[i] = 8035 + (int64_t)(xs64(x) % 2557); break;
            case 5: for (uint64_t i = 0; i < n; i++) {
                        uint64_t r = xs64(x) % 4;
                        rf[i] = (int8_t)(r == 0 ? 0 : (r == 1 ? 1 : 2));
                    }
                    break;
            case 6: for (uint64_t i = 0; i < n; i++) ls[i] = (int8_t)(xs64(x) % 2); break;
        }
    }
}

template <typename Acc, typename Fold>
void expected_scan(uint64_t n_rows, uint64_t seed, Acc &out, Fold fold)
{
    const uint64_t seed0 = seed ? seed : 42;
    const uint64_t BATCH = 1u << 20;
    const uint64_t n_batches = (n_rows + BATCH - 1) / BATCH;
    int threads = (int)std::thread::hardware_concurrency();
    if (threads <= 0) threads = 8;
    if ((uint64_t)threads > n_batches) threads = (int)n_batches;
    if (threads == 0) return;
    std::vector<Acc> parts(threads);
    std::atomic<uint64_t> next{0};
    auto worker = [&](int ti) {
        std::vector<int64_t> qty(BATCH), price(BATCH), disc(BATCH), tax(BATCH), ship(BATCH);
        std::vector<int8_t> rf(BATCH), ls(BATCH);
        for (;;) {
            uint64_t b = next.fetch_add(1);
            if (b >= n_batches) return;
            uint64_t n = n_rows - b * BATCH < BATCH ? n_rows - b * BATCH : BATCH;
            gen_batch(seed0, b, n, qty, price, disc, tax, ship, rf, ls);
            fold(parts[ti], n, qty.data(), price.data(), disc.data(),
                 tax.data(), ship.data(), rf.data(), ls.data());
        }
    };
    std::vector<std::thread> ts;
    for (int i = 0; i < threads; i++) ts.emplace_back(worker, i);
    for (auto &t : ts) t.join();
    for (auto &p : parts) out.merge(p);
}

} /* namespace */

/* Q6: sum(price*disc), count WHERE ship in [8766,9131) AND disc in [5,7]
 * AND qty < 2400 — the bench's fixed TPC-H Q6 shape */
extern "C" int csbench_expected_q6(uint64_t n_rows, uint64_t seed,
                                   int64_t *rev_lo, int64_t *rev_hi,
                                   int64_t *count)
{
    struct A : ExpQ6 {
        void merge(const A &o) { revenue += o.revenue; count += o.count; }
    } acc;
    expected_scan(n_rows, seed, acc,
        [](A &a, uint64_t n, const int64_t *qty, const int64_t *price,
           const int64_t *disc, const int64_t *tax, const int64_t *ship,
           const int8_t *, const int8_t *) {
            (void)tax;
            for (uint64_t i = 0; i < n; i++) {
                if (ship[i] >= 8766 && ship[i] < 9131 &&
                    disc[i] >= 5 && disc[i] <= 7 && qty[i] < 2400) {
                    a.revenue += (__int128)price[i] * disc[i];
                    a.count++;
                }
            }
        });
    *rev_lo = (int64_t)(uint64_t)acc.revenue;
    *rev_hi = (int64_t)(acc.revenue >> 64);
    *count = acc.count;
    return CSTRIPE_OK;
}

/* Q1: per (returnflag, linestatus): sum(qty), sum(price),
 * sum(price*(100-disc)), sum(price*(100-disc)*(100+tax)), count
 * WHERE ship <= 10471. Outputs indexed [rf*2+ls]. */
extern "C" int csbench_expected_q1(uint64_t n_rows, uint64_t seed,
                                   int64_t *lo, int64_t *hi, int64_t *counts)
{
    struct A : ExpQ1 {
        void merge(const A &o) {
            for (int g = 0; g < 6; g++) {
                sum_qty[g] += o.sum_qty[g];
                sum_price[g] += o.sum_price[g];
                sum_disc_price[g] += o.sum_disc_price[g];
                sum_charge[g] += o.sum_charge[g];
                count[g] += o.count[g];
            }
        }
    } acc;
    expected_scan(n_rows, seed, acc,
        [](A &a, uint64_t n, const int64_t *qty, const int64_t *price,
           const int64_t *disc, const int64_t *tax, const int64_t *ship,
           const int8_t *rf, const int8_t *ls) {
            for (uint64_t i = 0; i < n; i++) {
                if (ship[i] > 10471) continue;
                const int g = (int)rf[i] * 2 + (int)ls[i];
                a.sum_qty[g] += qty[i];
                a.sum_price[g] += price[i];
                __int128 dp = (__int128)price[i] * (100 - disc[i]);
                a.sum_disc_price[g] += dp;
                a.sum_charge[g] += dp * (100 + tax[i]);
                a.count[g]++;
            }
        });
    /* lo/hi laid out [group][4 sums] */
    for (int g = 0; g < 6; g++) {
        const __int128 v[4] = {acc.sum_qty[g], acc.sum_price[g],
                               acc.sum_disc_price[g], acc.sum_charge[g]};
        for (int k = 0; k < 4; k++) {
            lo[g * 4 + k] = (int64_t)(uint64_t)v[k];
            hi[g * 4 + k] = (int64_t)(v[k] >> 64);
        }
        counts[g] = acc.count[g];
    }
    return CSTRIPE_OK;
}
