/*
 * cstripe_host.cpp — host side of the MI355X-native columnar scan path:
 * format producer (writer), flat-file stripe directory (reader/footer),
 * chunk-group min/max pruning, and the combine surface.
 *
 * Reference semantics restated (never copied) from:
 *   writer:  columnar_writer.c  ColumnarWriteRow :168-266, FlushStripe
 *            :391-516, SerializeBoolArray :523-545, SerializeSingleDatum
 *            :555-585, SerializeChunkData :592-654, UpdateChunkSkipNodeMinMax
 *            :663-718
 *   codec:   columnar_compression.c CompressBuffer :62-158 (lz4 :71-94,
 *            zstd :97-121), DecompressBuffer :165-270
 *   pruning: columnar_reader.c SelectedChunkMask :1132-1187 (range
 *            refutation per qual var; count a chunk once)
 *   combine: distributed/utils/aggregate_utils.c coord_combine_agg core
 *            :820-1021 (strict: skip NULL partials, first non-NULL
 *            initializes); COUNT NULL->0 COALESCE
 *            multi_logical_optimizer.c:1831-1885
 */
#include "internal.h"
#include "compress.h"
#include "lz4_enc.h"
#include "pglz.h"
#include "zstd_r.h"

#include <hip/hip_runtime.h>

#include <cstdarg>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <algorithm>

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <dirent.h>
#include <string>
#include <unistd.h>

#ifdef _OPENMP
#include <omp.h>
#endif

/* ============================ errors ============================ */

static thread_local char g_errbuf[512] = "";

void cs_set_err(const char *fmt, ...)
{
    va_list ap;
    va_start(ap, fmt);
    vsnprintf(g_errbuf, sizeof(g_errbuf), fmt, ap);
    va_end(ap);
}

extern "C" const char *cstripe_errmsg(void) { return g_errbuf; }
extern "C" uint32_t cstripe_abi_version(void) { return CSTRIPE_ABI_VERSION; }

extern "C" void cstripe_default_options(cstripe_options *o)
{
    /* defaults mirror columnar.c:30-44 (stripe 150000, chunk 10000, level 3;
     * default codec here is LZ4 — the GPU decode path's native codec) */
    o->stripe_row_limit = 150000;
    o->chunk_group_row_limit = 10000;
    o->compression = CSTRIPE_COMP_LZ4;
    o->compression_level = 3;
    o->lz4_seg_target_kb = 0;
    o->lz4_seg_target_bytes = 256;
    o->lz4_min_match = 4;              /* standard greedy LZ4 parse */
    o->canonical = 1;                  /* closed-form canonical parses on */
}

static uint32_t type_width(uint8_t t) { return csf_type_width(t); }

/* PG float ordering (backend/utils/adt/float.c float8_cmp_internal):
 * NaN sorts greater than every non-NaN and equal to itself. Used for
 * writer min/max skip nodes and for chunk refutation so float pruning
 * matches the reference's operator semantics (round-1 advisor finding). */
static inline int f64_cmp_pg(double a, double b)
{
    if (a > b) return 1;
    if (a < b) return -1;
    if (a == b) return 0;
    const bool na = (a != a), nb = (b != b);
    if (na && nb) return 0;
    return na ? 1 : -1;
}

/* ============================ writer ============================ */

namespace {

struct ChunkOut {
    std::vector<uint8_t> exists_packed;
    std::vector<uint8_t> value_comp;      /* compressed (or raw) value stream */
    std::vector<csf_seg> segs;
    csf_skipnode node;                    /* offsets filled at flush */
};

struct ColCur {
    std::vector<uint8_t> values;          /* packed present values, current chunk */
    std::vector<uint8_t> exists;          /* 1 byte per row, current chunk */
    bool has_min_max = false;
    int64_t min_i = 0, max_i = 0;         /* physical encoding (f64 via bit pattern) */
    std::vector<uint8_t> raw_pending;     /* finalized raw value streams, pending compress */
};

struct PendingChunk {                     /* raw chunk awaiting compression at flush */
    uint32_t col;
    uint32_t chunk;
    std::vector<uint8_t> raw_values;
    std::vector<uint8_t> exists_packed;
    csf_skipnode node;                    /* min/max/row_count/n_present/decomp filled */
    /* device write path: stream already compressed on the GPU */
    bool has_ready = false;
    std::vector<uint8_t> ready_comp;
    uint8_t ready_mode = 0;
};

} /* namespace */

struct cstripe_writer {
    FILE *f = nullptr;
    std::string path;
    std::vector<csf_coldef> cols;
    cstripe_options opts{};
    uint64_t file_pos = 0;
    uint64_t total_rows = 0;

    /* current stripe */
    uint64_t stripe_first_row = 0;
    uint32_t stripe_rows = 0;             /* rows accumulated in current stripe */
    std::vector<ColCur> cur;              /* per column, current chunk state */
    std::vector<PendingChunk> pending;    /* finalized raw chunks of current stripe */
    std::vector<uint32_t> chunk_rows;     /* rows per finalized chunk */

    /* footer accumulation */
    std::vector<cs_stripe_info> stripes;
};

/* typed min/max update over a span of present values */
static void update_minmax(ColCur &c, uint8_t type, const uint8_t *vals, size_t n)
{
    if (n == 0) return;
    switch (type) {
        case CSTRIPE_I8: {
            const int8_t *v = (const int8_t *)vals;
            int8_t mn = v[0], mx = v[0];
            for (size_t i = 1; i < n; i++) { if (v[i] < mn) mn = v[i]; if (v[i] > mx) mx = v[i]; }
            if (!c.has_min_max) { c.min_i = mn; c.max_i = mx; c.has_min_max = true; }
            else { if (mn < (int8_t)c.min_i) c.min_i = mn; if (mx > (int8_t)c.max_i) c.max_i = mx; }
            break;
        }
        case CSTRIPE_I16: {
            const int16_t *v = (const int16_t *)vals;
            int16_t mn = v[0], mx = v[0];
            for (size_t i = 1; i < n; i++) { if (v[i] < mn) mn = v[i]; if (v[i] > mx) mx = v[i]; }
            if (!c.has_min_max) { c.min_i = mn; c.max_i = mx; c.has_min_max = true; }
            else { if (mn < (int16_t)c.min_i) c.min_i = mn; if (mx > (int16_t)c.max_i) c.max_i = mx; }
            break;
        }
        case CSTRIPE_I32: {
            const int32_t *v = (const int32_t *)vals;
            int32_t mn = v[0], mx = v[0];
            for (size_t i = 1; i < n; i++) { if (v[i] < mn) mn = v[i]; if (v[i] > mx) mx = v[i]; }
            if (!c.has_min_max) { c.min_i = mn; c.max_i = mx; c.has_min_max = true; }
            else { if (mn < (int32_t)c.min_i) c.min_i = mn; if (mx > (int32_t)c.max_i) c.max_i = mx; }
            break;
        }
        case CSTRIPE_I64: {
            const int64_t *v = (const int64_t *)vals;
            int64_t mn = v[0], mx = v[0];
            for (size_t i = 1; i < n; i++) { if (v[i] < mn) mn = v[i]; if (v[i] > mx) mx = v[i]; }
            if (!c.has_min_max) { c.min_i = mn; c.max_i = mx; c.has_min_max = true; }
            else { if (mn < c.min_i) c.min_i = mn; if (mx > c.max_i) c.max_i = mx; }
            break;
        }
        case CSTRIPE_F32: {
            const float *v = (const float *)vals;
            double mn = v[0], mx = v[0];
            for (size_t i = 1; i < n; i++) {
                if (f64_cmp_pg(v[i], mn) < 0) mn = v[i];
                if (f64_cmp_pg(v[i], mx) > 0) mx = v[i];
            }
            if (!c.has_min_max) { memcpy(&c.min_i, &mn, 8); memcpy(&c.max_i, &mx, 8); c.has_min_max = true; }
            else {
                double omn, omx; memcpy(&omn, &c.min_i, 8); memcpy(&omx, &c.max_i, 8);
                if (f64_cmp_pg(mn, omn) < 0) memcpy(&c.min_i, &mn, 8);
                if (f64_cmp_pg(mx, omx) > 0) memcpy(&c.max_i, &mx, 8);
            }
            break;
        }
        case CSTRIPE_F64: {
            const double *v = (const double *)vals;
            double mn = v[0], mx = v[0];
            for (size_t i = 1; i < n; i++) {
                if (f64_cmp_pg(v[i], mn) < 0) mn = v[i];
                if (f64_cmp_pg(v[i], mx) > 0) mx = v[i];
            }
            if (!c.has_min_max) { memcpy(&c.min_i, &mn, 8); memcpy(&c.max_i, &mx, 8); c.has_min_max = true; }
            else {
                double omn, omx; memcpy(&omn, &c.min_i, 8); memcpy(&omx, &c.max_i, 8);
                if (f64_cmp_pg(mn, omn) < 0) memcpy(&c.min_i, &mn, 8);
                if (f64_cmp_pg(mx, omx) > 0) memcpy(&c.max_i, &mx, 8);
            }
            break;
        }
        case CSTRIPE_TEXT: {
            /* min/max in the C-collation lex-key order (format.h
             * csf_text_lex_key) — used for EQ/NE refutation only */
            const uint32_t *v = (const uint32_t *)vals;
            int64_t mn = csf_text_lex_key(v[0]), mx = mn;
            for (size_t i = 1; i < n; i++) {
                const int64_t k = csf_text_lex_key(v[i]);
                if (k < mn) mn = k;
                if (k > mx) mx = k;
            }
            if (!c.has_min_max) { c.min_i = mn; c.max_i = mx; c.has_min_max = true; }
            else { if (mn < c.min_i) c.min_i = mn; if (mx > c.max_i) c.max_i = mx; }
            break;
        }
    }
}

/* SerializeBoolArray (columnar_writer.c:523-545): bit i of byte i/8, LSB first */
static std::vector<uint8_t> pack_bools(const std::vector<uint8_t> &b)
{
    size_t n = b.size();
    std::vector<uint8_t> out((n + 7) / 8, 0);
    for (size_t i = 0; i < n; i++)
        if (b[i]) out[i / 8] |= (uint8_t)(1u << (i % 8));
    return out;
}

extern "C" cstripe_writer *cstripe_write_begin(const char *path, const cstripe_coldef *cols,
                                               uint32_t n_cols, const cstripe_options *opts)
{
    if (!path || !cols || n_cols == 0 || n_cols > 64) { cs_set_err("write_begin: bad args"); return nullptr; }
    FILE *f = fopen(path, "wb");
    if (!f) { cs_set_err("write_begin: cannot open %s", path); return nullptr; }

    auto *w = new cstripe_writer();
    w->f = f;
    w->path = path;
    if (opts) w->opts = *opts; else cstripe_default_options(&w->opts);
    if (w->opts.lz4_seg_target_bytes == 0 && w->opts.lz4_seg_target_kb == 0)
        w->opts.lz4_seg_target_bytes = 256;
    /* segment decomp_len carries the stream-shape mode in bits 24-31, so a
     * single-segment chunk's raw stream must fit 24 bits (default 10000-row
     * chunks are 80 KB; 1M rows x 8 B = 8 MB still fits with margin) */
    if ((uint64_t)w->opts.chunk_group_row_limit * 8 >= (1u << 24) ||
        w->opts.chunk_group_row_limit == 0) {
        cs_set_err("chunk_group_row_limit %u out of range (1 .. 2097151)",
                   w->opts.chunk_group_row_limit);
        fclose(f);
        delete w;
        return nullptr;
    }
    for (uint32_t i = 0; i < n_cols; i++) {
        csf_coldef d{};
        memcpy(d.name, cols[i].name, sizeof(d.name));
        d.name[31] = 0;
        d.type = cols[i].type;
        d.scale = cols[i].scale;
        if (type_width(d.type) == 0) { cs_set_err("write_begin: bad column type %d", d.type); fclose(f); delete w; return nullptr; }
        w->cols.push_back(d);
    }
    w->cur.resize(n_cols);

    uint8_t hdr[CSF_HEADER_SIZE] = {0};
    memcpy(hdr, CSF_MAGIC, 8);
    uint32_t ver = CSF_VERSION;
    memcpy(hdr + 8, &ver, 4);
    fwrite(hdr, 1, CSF_HEADER_SIZE, f);
    w->file_pos = CSF_HEADER_SIZE;
    return w;
}

/* finalize the current chunk of every column into w->pending (raw; compression
 * deferred to flush so it can run in parallel across chunks) */
static void finalize_chunk(cstripe_writer *w, uint32_t rows_in_chunk)
{
    uint32_t chunk_index = (uint32_t)w->chunk_rows.size();
    for (uint32_t c = 0; c < w->cols.size(); c++) {
        ColCur &cc = w->cur[c];
        PendingChunk pc;
        pc.col = c;
        pc.chunk = chunk_index;
        pc.raw_values = std::move(cc.values);
        pc.exists_packed = pack_bools(cc.exists);
        memset(&pc.node, 0, sizeof(pc.node));
        pc.node.min_i = cc.min_i;
        pc.node.max_i = cc.max_i;
        pc.node.has_min_max = cc.has_min_max ? 1 : 0;
        pc.node.row_count = rows_in_chunk;
        uint32_t width = type_width(w->cols[c].type);
        pc.node.n_present = (uint32_t)(pc.raw_values.size() / width);
        pc.node.decompressed_size = pc.raw_values.size();
        w->pending.push_back(std::move(pc));
        cc.values.clear();
        cc.exists.clear();
        cc.has_min_max = false;
        cc.min_i = cc.max_i = 0;
    }
    w->chunk_rows.push_back(rows_in_chunk);
}

/* compress one pending chunk's value stream into out (segments); mirrors
 * SerializeChunkData/CompressBuffer: lz4/zstd per segment, raw when the codec
 * is NONE or fails. Returns false only on a hard format-capacity error
 * (> CSF segment cap), never on incompressible data. */
static bool compress_chunk(const cstripe_writer *w, const PendingChunk &pc, ChunkOut &out)
{
    out.node = pc.node;
    out.exists_packed = pc.exists_packed;
    if (pc.has_ready) {                   /* GPU-compressed canonical stream */
        out.value_comp = pc.ready_comp;
        csf_seg s{0, (uint32_t)pc.ready_comp.size(), 0,
                  (uint32_t)pc.node.decompressed_size |
                  ((uint32_t)pc.ready_mode << 24)};
        out.segs.push_back(s);
        out.node.comp_type = CSTRIPE_COMP_LZ4;
        out.node.comp_level = 0;
        out.node.n_segs = 1;
        out.node.value_len = pc.ready_comp.size();
        return true;
    }
    const std::vector<uint8_t> &raw = pc.raw_values;
    uint8_t codec = w->opts.compression;
    size_t target = w->opts.lz4_seg_target_bytes ? w->opts.lz4_seg_target_bytes
                                                 : (size_t)w->opts.lz4_seg_target_kb * 1024;
    uint32_t width = type_width(w->cols[pc.col].type);

    if (codec == CSTRIPE_COMP_PGLZ && !raw.empty()) {
        /* reference PG_LZ layout: ColumnarCompressHeader (varlena len +
         * rawsize) then the pglz stream, one whole-chunk buffer
         * (columnar_compression.c:122-151) */
        out.value_comp.resize(raw.size() + CSPGLZ_HDRSZ);
        int32_t csz = cspglz_compress(raw.data(), (int32_t)raw.size(),
                                      out.value_comp.data() + CSPGLZ_HDRSZ,
                                      (int32_t)raw.size());
        if (csz > 0) {
            out.value_comp.resize((size_t)csz + CSPGLZ_HDRSZ);
            cspglz_set_header(out.value_comp.data(),
                              (uint32_t)out.value_comp.size(),
                              (int32_t)raw.size());
            csf_seg s{0, (uint32_t)out.value_comp.size(), 0, (uint32_t)raw.size()};
            out.segs.push_back(s);
            out.node.comp_type = CSTRIPE_COMP_PGLZ;
            out.node.comp_level = 0;
            out.node.n_segs = 1;
            out.node.value_len = out.value_comp.size();
            return true;
        }
        out.value_comp.clear();
        codec = CSTRIPE_COMP_NONE;      /* did not shrink -> raw */
    }
    if (raw.empty() || codec == CSTRIPE_COMP_NONE) {
        out.value_comp = raw;
        csf_seg s{0, (uint32_t)raw.size(), 0, (uint32_t)raw.size()};
        out.segs.push_back(s);
        out.node.comp_type = CSTRIPE_COMP_NONE;
        out.node.comp_level = 0;
        out.node.n_segs = 1;
        out.node.value_len = raw.size();
        return true;
    }

    /* ---- canonical parses (closed-form GPU access; see format.h) ----
     * The emitted stream is a standard LZ4 block / zstd frame either way;
     * canonical just picks a parse whose value positions are computable,
     * killing the GPU's per-sequence/FSE parse chain (round-1's instruction
     * wall). */
    if ((codec == CSTRIPE_COMP_LZ4 || codec == CSTRIPE_COMP_ZSTD) &&
        w->opts.canonical && width == 8 &&
        raw.size() >= 24 && raw.size() < (1u << 24) &&
        (codec == CSTRIPE_COMP_LZ4 || raw.size() < (127u << 10))) {
        const size_t nv = raw.size() / 8;
        uint64_t v0, x = 0;
        memcpy(&v0, raw.data(), 8);
        const uint8_t *rp = raw.data();
        for (size_t i = 1; i < nv; i++) {
            uint64_t vi;
            memcpy(&vi, rp + i * 8, 8);
            x |= vi ^ v0;
        }
        static const zr_ctables zct_c = [] {
            zr_ctables t;
            zr_build_ctables(&t);
            return t;
        }();
        int mode = -1, csz = 0;
        size_t cap = raw.size() + raw.size() / 8 + 64;
        out.value_comp.resize(cap);
        if (x == 0) {
            if (codec == CSTRIPE_COMP_LZ4) {
                csz = lz4e_canon_const(rp, (int)nv, out.value_comp.data(), (int)cap);
                if (csz > 0) mode = CSF_SEGMODE_CONST;
            } else {
                csz = zr_canon_const(rp, (int)nv, out.value_comp.data(), (int)cap, &zct_c);
                if (csz > 0) mode = CSF_SEGMODE_ZR_CONST;
            }
        } else {
            int L = (63 - __builtin_clzll(x)) / 8 + 1;
            if (L <= 4) {
                if (codec == CSTRIPE_COMP_LZ4) {
                    csz = lz4e_canon_p(rp, (int)nv, L, out.value_comp.data(), (int)cap);
                    if (csz > 0) mode = (int)(CSF_SEGMODE_P_BASE | (uint32_t)L);
                } else {
                    csz = zr_canon_p(rp, (int)nv, L, out.value_comp.data(), (int)cap, &zct_c);
                    if (csz > 0) mode = (int)(CSF_SEGMODE_ZRP_BASE | (uint32_t)L);
                }
            }
            /* L > 4: high bytes vary — no canonical parse; the greedy path
             * below runs, and its incompressible->NONE fallback (reference
             * CompressBuffer semantics) already yields closed-form access */
        }
        if (mode >= 0 && (size_t)csz >= raw.size())
            mode = -1;   /* did not shrink -> raw NONE below (reference rule) */
        if (mode >= 0) {
            out.value_comp.resize((size_t)csz);
            csf_seg s{0, (uint32_t)csz, 0,
                      (uint32_t)raw.size() | ((uint32_t)mode << 24)};
            out.segs.push_back(s);
            out.node.comp_type = codec;
            out.node.comp_level = 0;
            out.node.n_segs = 1;
            out.node.value_len = (uint64_t)csz;
            return true;
        }
    }

    /* width-4 canonical zstd: slots that differ in at most ONE byte (the
     * char(1) short-varlena case: [0x05 ch 00 00] varies only in byte 1) —
     * expressible in zstd (min match 3) but not LZ4 (min match 4); the GPU
     * reads one byte per row closed-form (csf_canon_zr4b_pos) */
    if (codec == CSTRIPE_COMP_ZSTD && w->opts.canonical && width == 4 &&
        raw.size() >= 12 && raw.size() < (127u << 10)) {
        const size_t nv = raw.size() / 4;
        uint32_t v0, x = 0;
        memcpy(&v0, raw.data(), 4);
        const uint8_t *rp = raw.data();
        for (size_t i = 1; i < nv; i++) {
            uint32_t vi;
            memcpy(&vi, rp + i * 4, 4);
            x |= vi ^ v0;
        }
        static const zr_ctables zct4 = [] {
            zr_ctables t;
            zr_build_ctables(&t);
            return t;
        }();
        int mode = -1, csz = 0, k = -1;
        if (x == 0) {
            k = 4;                           /* const form */
        } else {
            for (int b = 0; b < 4; b++)
                if ((x & ~(0xFFu << (8 * b))) == 0) { k = b; break; }
        }
        if (k >= 0) {
            size_t cap = raw.size() + 64;
            out.value_comp.resize(cap);
            if (k == 4) {
                csz = zr_canon_const4(rp, (int)nv, out.value_comp.data(), (int)cap, &zct4);
                if (csz > 0) mode = CSF_SEGMODE_ZR4_CONST;
            } else {
                csz = zr_canon_b4(rp, (int)nv, k, out.value_comp.data(), (int)cap, &zct4);
                if (csz > 0) mode = (int)(CSF_SEGMODE_ZR4B_BASE | (uint32_t)k);
            }
        }
        if (mode >= 0 && (size_t)csz >= raw.size())
            mode = -1;
        if (mode >= 0) {
            out.value_comp.resize((size_t)csz);
            csf_seg s{0, (uint32_t)csz, 0,
                      (uint32_t)raw.size() | ((uint32_t)mode << 24)};
            out.segs.push_back(s);
            out.node.comp_type = codec;
            out.node.comp_level = 0;
            out.node.n_segs = 1;
            out.node.value_len = (uint64_t)csz;
            return true;
        }
        out.value_comp.clear();
    }

    /* segment boundaries 16 B-aligned (pure byte split of the raw stream;
     * lets the GPU flush decoded segments with aligned 16 B stores) */
    (void)width;
    size_t n = raw.size();
    size_t per = (target / 16) * 16;    /* exact target (16B-aligned splits) —
                                         * uniform segments are what the fused
                                         * GPU kernel's tile math assumes */
    if (codec == CSTRIPE_COMP_ZSTD && per < 256)
        per = 256;                      /* ~13 B frame overhead (5%) is worth
                                         * the GPU decoder occupancy: LDS
                                         * bytes/frame bound waves/CU, and
                                         * 256 B frames double them vs 512 */
    if (per == 0) per = n;
    /* caps: segment decomp_len carries the mode byte in bits 24-31 (8 MB is
     * comfortably under 2^24), and a restricted-zstd RAW-block fallback is
     * bounded by the 21-bit block size (1 MB) */
    if (per > (1u << 23)) per = 1u << 23;
    if (codec == CSTRIPE_COMP_ZSTD && per > (96u << 10))
        per = 96u << 10;               /* zstd block content cap is 128 KB */

    out.value_comp.clear();
    out.segs.clear();
    size_t off = 0;
    bool ok = true;
    lz4e_state enc;                   /* one init per chunk, reused across
                                       * segments (see lz4_enc.h) */
    if (codec == CSTRIPE_COMP_LZ4) lz4e_init(&enc);
    while (off < n) {
        size_t len = std::min(per, n - off);
        /* absorb a tiny (<16 B) tail only in large-segment mode: the GPU
         * fused/lane kernels assume non-final segments are EXACTLY `per`
         * decompressed bytes and the final one is <= per, so for the
         * 256 B lane-parallel default a 257-271 B absorbed final segment
         * would be read from the wrong lane region (round-1 advisor
         * finding); a standalone <16 B final segment decodes fine */
        if (per > 256 && n - off - len < 16) len = n - off;
        csf_seg s{};
        s.decomp_off = (uint32_t)off;
        s.decomp_len = (uint32_t)len;
        s.comp_off = (uint32_t)out.value_comp.size();
        if (codec == CSTRIPE_COMP_LZ4) {
            int bound = LZ4_compressBound((int)len);
            size_t base = out.value_comp.size();
            out.value_comp.resize(base + (size_t)bound);
            /* width-granular matches for 4-byte slot streams (TEXT/I32/F32):
             * the GPU lane decoder's aligned-word path then covers every copy */
            const int malign = (width == 4) ? 4 : 1;
            int csz = lz4e_compress_mm_a(&enc, raw.data() + off,
                                         (int)len, out.value_comp.data() + base, bound,
                                         w->opts.lz4_min_match >= 4 ? w->opts.lz4_min_match : 4,
                                         malign);
            if (csz <= 0)             /* capacity fallback: system liblz4 */
                csz = LZ4_compress_default((const char *)raw.data() + off,
                                           (char *)out.value_comp.data() + base,
                                           (int)len, bound);
            if (csz <= 0) { ok = false; break; }
            out.value_comp.resize(base + (size_t)csz);
            s.comp_len = (uint32_t)csz;
        } else if (codec == CSTRIPE_COMP_ZSTD) {
            /* restricted zstd frames (zstd_r.h): standard RFC 8878 frames
             * the GPU lane decoder handles (raw literals + predefined-FSE
             * sequences); libzstd-decodable, pinned in tests. Falls back to
             * a raw-block frame when the parse does not shrink a segment. */
            /* magic static: thread-safe one-time build (C++11) */
            static const zr_ctables zct = [] {
                zr_ctables t;
                zr_build_ctables(&t);
                return t;
            }();
            size_t bound = (size_t)len + len / 2 + 64;
            size_t base = out.value_comp.size();
            out.value_comp.resize(base + bound);
            int csz = zr_compress(raw.data() + off, (int)len,
                                  out.value_comp.data() + base, (int)bound, &zct);
            if (csz <= 0) {
                /* raw-block frame: magic+fhd+fcs+bh then the bytes */
                uint8_t *d = out.value_comp.data() + base;
                int op2 = 0;
                d[op2++] = 0x28; d[op2++] = 0xB5; d[op2++] = 0x2F; d[op2++] = 0xFD;
                if (len <= 255) { d[op2++] = 0x20; d[op2++] = (uint8_t)len; }
                else if (len <= 65535 + 256) {
                    d[op2++] = 0x60;
                    uint32_t f2 = (uint32_t)len - 256;
                    d[op2++] = (uint8_t)f2; d[op2++] = (uint8_t)(f2 >> 8);
                } else {
                    d[op2++] = 0xA0;
                    uint32_t f2 = (uint32_t)len;
                    d[op2++] = (uint8_t)f2; d[op2++] = (uint8_t)(f2 >> 8);
                    d[op2++] = (uint8_t)(f2 >> 16); d[op2++] = (uint8_t)(f2 >> 24);
                }
                uint32_t bh = 1u | (0u << 1) | ((uint32_t)len << 3);
                d[op2++] = (uint8_t)bh; d[op2++] = (uint8_t)(bh >> 8); d[op2++] = (uint8_t)(bh >> 16);
                memcpy(d + op2, raw.data() + off, len);
                csz = op2 + (int)len;
            }
            out.value_comp.resize(base + (size_t)csz);
            s.comp_len = (uint32_t)csz;
            s.decomp_len |= CSF_SEGMODE_ZR << 24;
        } else {
            ok = false; break;
        }
        out.segs.push_back(s);
        off += len;
    }

    /* incompressible -> keep raw, like the reference (CompressBuffer returns
     * false when compression does not shrink, columnar_compression.c:88-94;
     * raw NONE chunks also get closed-form GPU access for free) */
    if (ok && out.value_comp.size() >= raw.size())
        ok = false;
    if (!ok) {  /* CompressBuffer returned false -> keep uncompressed */
        out.value_comp = raw;
        out.segs.clear();
        csf_seg s{0, (uint32_t)raw.size(), 0, (uint32_t)raw.size()};
        out.segs.push_back(s);
        out.node.comp_type = CSTRIPE_COMP_NONE;
        out.node.comp_level = 0;
    } else {
        out.node.comp_type = codec;
        out.node.comp_level = w->opts.compression_level;
    }
    /* reader cap is 4096 segments and n_segs is uint16 — refuse to emit a
     * chunk the reader would reject (or, after wraparound, misparse) instead
     * of silently truncating (round-1 advisor finding) */
    if (out.segs.size() > 4096) {
        cs_set_err("chunk needs %zu segments (> 4096 cap): raise "
                   "lz4_seg_target_bytes or lower chunk_group_row_limit",
                   out.segs.size());
        return false;
    }
    out.node.n_segs = (uint16_t)out.segs.size();
    out.node.value_len = out.value_comp.size();
    return true;
}

/* FlushStripe (columnar_writer.c:391-516): compress pending chunks, lay out
 * per column all exists streams then all value streams, offsets relative to
 * stripe data start; append stripe metadata for the footer */
static int flush_stripe(cstripe_writer *w)
{
    if (w->stripe_rows == 0) return CSTRIPE_OK;
    /* finalize a partial last chunk (FlushStripe :419-422) */
    uint32_t done_rows = 0;
    for (uint32_t r : w->chunk_rows) done_rows += r;
    if (done_rows < w->stripe_rows)
        finalize_chunk(w, w->stripe_rows - done_rows);

    uint32_t n_cols = (uint32_t)w->cols.size();
    uint32_t n_chunks = (uint32_t)w->chunk_rows.size();

    /* compress all pending chunks (parallel; independent per column-chunk) */
    std::vector<ChunkOut> outs(w->pending.size());
    bool comp_ok = true;
    #pragma omp parallel for schedule(dynamic)
    for (long i = 0; i < (long)w->pending.size(); i++)
        if (!compress_chunk(w, w->pending[i], outs[i]))
            comp_ok = false;
    if (!comp_ok) {
        /* compress_chunk set the detailed message on its OMP worker thread's
         * buffer; restate on the caller's thread where errmsg() reads */
        cs_set_err("chunk exceeds the 4096-segment cap: raise "
                   "lz4_seg_target_bytes or lower chunk_group_row_limit");
        return CSTRIPE_ERR_FORMAT;
    }

    /* outs indexed by pending order: chunk-major (all columns of chunk 0, then
     * chunk 1, ...). Reorder access as [col][chunk]. */
    auto out_at = [&](uint32_t col, uint32_t chunk) -> ChunkOut & {
        return outs[(size_t)chunk * n_cols + col];
    };

    /* compute offsets: per column, exists streams then value streams
     * (FlushStripe :425-458) */
    uint64_t stripe_size = 0;
    for (uint32_t c = 0; c < n_cols; c++) {
        for (uint32_t k = 0; k < n_chunks; k++) {
            ChunkOut &o = out_at(c, k);
            o.node.exists_off = stripe_size;
            o.node.exists_len = o.exists_packed.size();
            stripe_size += o.exists_packed.size();
        }
        for (uint32_t k = 0; k < n_chunks; k++) {
            ChunkOut &o = out_at(c, k);
            o.node.value_off = stripe_size;
            stripe_size += o.node.value_len;
        }
    }

    uint64_t stripe_file_off = w->file_pos;

    /* write data: per column all exists, then all values (FlushStripe :477-502) */
    for (uint32_t c = 0; c < n_cols; c++) {
        for (uint32_t k = 0; k < n_chunks; k++) {
            ChunkOut &o = out_at(c, k);
            if (!o.exists_packed.empty() &&
                fwrite(o.exists_packed.data(), 1, o.exists_packed.size(), w->f) != o.exists_packed.size())
                { cs_set_err("flush: short write"); return CSTRIPE_ERR_IO; }
        }
        for (uint32_t k = 0; k < n_chunks; k++) {
            ChunkOut &o = out_at(c, k);
            if (!o.value_comp.empty() &&
                fwrite(o.value_comp.data(), 1, o.value_comp.size(), w->f) != o.value_comp.size())
                { cs_set_err("flush: short write"); return CSTRIPE_ERR_IO; }
        }
    }
    w->file_pos += stripe_size;

    /* record stripe for the footer */
    cs_stripe_info si;
    si.meta.file_offset = stripe_file_off;
    si.meta.data_size = stripe_size;
    si.meta.first_row_number = w->stripe_first_row;
    si.meta.row_count = w->stripe_rows;
    si.meta.chunk_count = n_chunks;
    si.meta.reserved = 0;
    si.group_rows = w->chunk_rows;
    si.nodes.resize(n_cols);
    for (uint32_t c = 0; c < n_cols; c++) {
        si.nodes[c].resize(n_chunks);
        for (uint32_t k = 0; k < n_chunks; k++) {
            si.nodes[c][k].n = out_at(c, k).node;
            si.nodes[c][k].segs = out_at(c, k).segs;
        }
    }
    w->stripes.push_back(std::move(si));

    /* reset stripe state */
    w->pending.clear();
    w->chunk_rows.clear();
    w->stripe_first_row += w->stripe_rows;
    w->stripe_rows = 0;
    return CSTRIPE_OK;
}

/* device-side append: full chunks compress on the GPU (csgpu_compress_chunk,
 * cstripe_gpu.hip); partial spans round-trip through the host accumulation
 * path to keep chunk/stripe boundary semantics identical */
extern "C" int cstripe_write_rows_device(cstripe_writer *w, uint64_t n_rows,
                                         const void *const *dev_values)
{
    if (!w || !dev_values) { cs_set_err("write_rows_device: bad args"); return CSTRIPE_ERR_ARG; }
    const uint32_t n_cols = (uint32_t)w->cols.size();
    const uint32_t chunk_limit = w->opts.chunk_group_row_limit;
    const uint64_t stripe_limit = w->opts.stripe_row_limit;
    std::vector<uint8_t> types(n_cols);
    for (uint32_t c = 0; c < n_cols; c++) types[c] = w->cols[c].type;

    uint64_t done = 0;
    while (done < n_rows) {
        const uint32_t in_chunk = (uint32_t)(w->stripe_rows % chunk_limit);
        const uint64_t to_chunk = chunk_limit - in_chunk;
        const uint64_t to_stripe = stripe_limit - w->stripe_rows;
        const uint64_t span = std::min({n_rows - done, to_chunk, to_stripe});

        const bool full_chunk = in_chunk == 0 && span == chunk_limit &&
                                w->opts.compression == CSTRIPE_COMP_LZ4 &&
                                w->opts.canonical;
        if (full_chunk) {
            std::vector<const void *> vp(n_cols);
            for (uint32_t c = 0; c < n_cols; c++) vp[c] = dev_values[c];
            std::vector<cs_dev_chunk_col> cc;
            int rc = csgpu_compress_chunk(vp.data(), types.data(), n_cols,
                                          (uint32_t)span, done, cc);
            if (rc != CSTRIPE_OK) return rc;
            const uint32_t chunk_index = (uint32_t)w->chunk_rows.size();
            std::vector<uint8_t> ones((span + 7) / 8, 0xFF);
            if (span % 8) ones.back() = (uint8_t)(0xFF >> (8 - span % 8));
            for (uint32_t c = 0; c < n_cols; c++) {
                PendingChunk pc;
                pc.col = c;
                pc.chunk = chunk_index;
                memset(&pc.node, 0, sizeof(pc.node));
                pc.node.min_i = cc[c].min_i;
                pc.node.max_i = cc[c].max_i;
                pc.node.has_min_max = cc[c].has_min_max ? 1 : 0;
                pc.node.row_count = span;
                pc.node.n_present = (uint32_t)span;
                pc.node.decompressed_size = span * type_width(types[c]);
                pc.exists_packed = ones;
                if (cc[c].canonical) {
                    pc.has_ready = true;
                    pc.ready_comp = std::move(cc[c].data);
                    pc.ready_mode = cc[c].mode;
                } else {
                    pc.raw_values = std::move(cc[c].data);
                }
                w->pending.push_back(std::move(pc));
            }
            w->chunk_rows.push_back((uint32_t)span);
            w->stripe_rows += (uint32_t)span;
            w->total_rows += span;
        } else {
            /* partial span / non-canonical codec: host accumulation path */
            std::vector<std::vector<uint8_t>> host(n_cols);
            std::vector<const void *> hp(n_cols);
            for (uint32_t c = 0; c < n_cols; c++) {
                const uint32_t width = type_width(types[c]);
                host[c].resize(span * width);
                if (hipMemcpy(host[c].data(),
                              (const uint8_t *)dev_values[c] + done * width,
                              span * width, hipMemcpyDeviceToHost) != hipSuccess) {
                    cs_set_err("write_rows_device: D2H copy failed");
                    return CSTRIPE_ERR;
                }
                hp[c] = host[c].data();
            }
            int rc = cstripe_write_rows(w, span, hp.data(), nullptr);
            if (rc != CSTRIPE_OK) return rc;
            done += span;
            continue;
        }
        done += span;
        if (w->stripe_rows >= stripe_limit) {
            int rc = flush_stripe(w);
            if (rc != CSTRIPE_OK) return rc;
        }
    }
    return CSTRIPE_OK;
}

extern "C" int cstripe_write_rows(cstripe_writer *w, uint64_t n_rows,
                                  const void *const *values, const uint8_t *const *nulls)
{
    if (!w || !values) { cs_set_err("write_rows: bad args"); return CSTRIPE_ERR_ARG; }
    uint32_t n_cols = (uint32_t)w->cols.size();
    const uint32_t chunk_limit = w->opts.chunk_group_row_limit;
    const uint64_t stripe_limit = w->opts.stripe_row_limit;

    uint64_t done = 0;
    while (done < n_rows) {
        uint32_t in_chunk = w->stripe_rows % chunk_limit;
        uint64_t to_chunk = chunk_limit - in_chunk;
        uint64_t to_stripe = stripe_limit - w->stripe_rows;
        uint64_t span = std::min({n_rows - done, to_chunk, to_stripe});

        for (uint32_t c = 0; c < n_cols; c++) {
            ColCur &cc = w->cur[c];
            uint32_t width = type_width(w->cols[c].type);
            const uint8_t *src = (const uint8_t *)values[c] + done * width;
            const uint8_t *nl = (nulls && nulls[c]) ? nulls[c] + done : nullptr;
            if (w->cols[c].type == CSTRIPE_TEXT) {
                /* slots must be well-formed short varlena (hdr odd, total
                 * length 2..4 incl. header: payload <= 3 bytes) */
                for (uint64_t i = 0; i < span; i++) {
                    if (nl && nl[i]) continue;
                    const uint8_t hdr = src[i * 4];
                    const uint32_t tot = hdr >> 1;
                    if (!(hdr & 1) || tot < 2 || tot > 4) {
                        cs_set_err("text slot row %llu: not a short varlena "
                                   "with payload <= 3 bytes (hdr 0x%02x)",
                                   (unsigned long long)(done + i), hdr);
                        return CSTRIPE_ERR_ARG;
                    }
                }
            }
            size_t base = cc.exists.size();
            cc.exists.resize(base + span);
            if (!nl) {
                memset(cc.exists.data() + base, 1, span);
                size_t vbase = cc.values.size();
                cc.values.resize(vbase + span * width);
                memcpy(cc.values.data() + vbase, src, span * width);
                update_minmax(cc, w->cols[c].type, src, span);
            } else {
                for (uint64_t i = 0; i < span; i++) {
                    if (nl[i]) { cc.exists[base + i] = 0; continue; }
                    cc.exists[base + i] = 1;
                    size_t vbase = cc.values.size();
                    cc.values.resize(vbase + width);
                    memcpy(cc.values.data() + vbase, src + i * width, width);
                    update_minmax(cc, w->cols[c].type, src + i * width, 1);
                }
            }
        }

        w->stripe_rows += (uint32_t)span;
        w->total_rows += span;
        done += span;

        if (w->stripe_rows % chunk_limit == 0)
            finalize_chunk(w, chunk_limit);
        if (w->stripe_rows >= stripe_limit) {
            int rc = flush_stripe(w);
            if (rc != CSTRIPE_OK) return rc;
        }
    }
    return CSTRIPE_OK;
}

static void footer_append(std::vector<uint8_t> &buf, const void *p, size_t n)
{
    const uint8_t *b = (const uint8_t *)p;
    buf.insert(buf.end(), b, b + n);
}

extern "C" int cstripe_write_end(cstripe_writer *w)
{
    if (!w) return CSTRIPE_ERR_ARG;
    int rc = flush_stripe(w);
    if (rc != CSTRIPE_OK) { cstripe_write_abort(w); return rc; }

    std::vector<uint8_t> fb;
    csf_footer_head head{};
    head.version = CSF_VERSION;
    head.column_count = (uint32_t)w->cols.size();
    head.stripe_count = (uint32_t)w->stripes.size();
    head.chunk_row_limit = w->opts.chunk_group_row_limit;
    head.stripe_row_limit = w->opts.stripe_row_limit;
    head.total_rows = w->total_rows;
    head.compression = w->opts.compression;
    head.compression_level = w->opts.compression_level;
    head.lz4_seg_target_kb = w->opts.lz4_seg_target_kb;
    footer_append(fb, &head, sizeof(head));
    for (auto &c : w->cols) footer_append(fb, &c, sizeof(c));
    for (auto &s : w->stripes) {
        footer_append(fb, &s.meta, sizeof(s.meta));
        footer_append(fb, s.group_rows.data(), s.group_rows.size() * 4);
        for (auto &colnodes : s.nodes) {
            for (auto &nd : colnodes) {
                footer_append(fb, &nd.n, sizeof(nd.n));
                footer_append(fb, nd.segs.data(), nd.segs.size() * sizeof(csf_seg));
            }
        }
    }
    uint64_t foff = w->file_pos;
    if (fwrite(fb.data(), 1, fb.size(), w->f) != fb.size()) { cs_set_err("footer: short write"); cstripe_write_abort(w); return CSTRIPE_ERR_IO; }
    fwrite(&foff, 1, 8, w->f);
    fwrite(CSF_FOOT_MAGIC, 1, 8, w->f);
    int frc = fclose(w->f);
    w->f = nullptr;
    delete w;
    if (frc != 0) { cs_set_err("close failed"); return CSTRIPE_ERR_IO; }
    return CSTRIPE_OK;
}

extern "C" void cstripe_write_abort(cstripe_writer *w)
{
    if (!w) return;
    if (w->f) fclose(w->f);
    delete w;
}

/* ============================ reader ============================ */

/* parse one shard file's footer, appending its stripes (tagged file_idx) */
static int footer_read_one(cstripe_reader *r, uint32_t file_idx)
{
    const cs_file &fl = r->files[file_idx];
    const uint8_t *map = fl.map;
    size_t map_size = fl.map_size;
    if (map_size < CSF_HEADER_SIZE + 16) { cs_set_err("file too small"); return CSTRIPE_ERR_FORMAT; }
    if (memcmp(map, CSF_MAGIC, 8) != 0) { cs_set_err("bad header magic"); return CSTRIPE_ERR_FORMAT; }
    if (memcmp(map + map_size - 8, CSF_FOOT_MAGIC, 8) != 0) { cs_set_err("bad footer magic"); return CSTRIPE_ERR_FORMAT; }
    uint64_t foff;
    memcpy(&foff, map + map_size - 16, 8);
    if (foff + 16 > map_size) { cs_set_err("bad footer offset"); return CSTRIPE_ERR_FORMAT; }

    const uint8_t *p = map + foff;
    const uint8_t *end = map + map_size - 16;
    auto need = [&](size_t n) -> bool { return (size_t)(end - p) >= n; };

    csf_footer_head head;
    if (!need(sizeof(csf_footer_head))) { cs_set_err("truncated footer"); return CSTRIPE_ERR_FORMAT; }
    memcpy(&head, p, sizeof(head));
    p += sizeof(head);
    if (head.version != CSF_VERSION) { cs_set_err("bad version %u", head.version); return CSTRIPE_ERR_FORMAT; }
    if (head.column_count == 0 || head.column_count > 64) { cs_set_err("bad column count"); return CSTRIPE_ERR_FORMAT; }

    std::vector<csf_coldef> cols(head.column_count);
    for (auto &c : cols) {
        if (!need(sizeof(c))) { cs_set_err("truncated coldefs"); return CSTRIPE_ERR_FORMAT; }
        memcpy(&c, p, sizeof(c));
        p += sizeof(c);
    }
    if (file_idx == 0) {
        r->head = head;
        r->cols = cols;
    } else {
        /* shard schema must match (same physical types/widths) */
        if (head.column_count != r->head.column_count) { cs_set_err("shard schema mismatch"); return CSTRIPE_ERR_FORMAT; }
        for (uint32_t i = 0; i < head.column_count; i++)
            if (cols[i].type != r->cols[i].type) { cs_set_err("shard schema mismatch (col %u)", i); return CSTRIPE_ERR_FORMAT; }
        /* scan tiling derives tiles_per_group from file 0's chunk_row_limit;
         * a shard with a larger limit would have rows past that bound
         * silently skipped (round-1 advisor finding) — reject mismatches */
        if (head.chunk_row_limit != r->head.chunk_row_limit) {
            cs_set_err("shard chunk_group_row_limit mismatch (%u vs %u)",
                       head.chunk_row_limit, r->head.chunk_row_limit);
            return CSTRIPE_ERR_FORMAT;
        }
        r->head.total_rows += head.total_rows;
        r->head.stripe_count += head.stripe_count;
    }

    size_t base = r->stripes.size();
    r->stripes.resize(base + head.stripe_count);
    for (size_t si = base; si < r->stripes.size(); si++) {
        auto &s = r->stripes[si];
        s.file_idx = file_idx;
        if (!need(sizeof(s.meta))) { cs_set_err("truncated stripe meta"); return CSTRIPE_ERR_FORMAT; }
        memcpy(&s.meta, p, sizeof(s.meta));
        p += sizeof(s.meta);
        if (s.meta.chunk_count > 1u << 22) { cs_set_err("bad chunk count"); return CSTRIPE_ERR_FORMAT; }
        s.group_rows.resize(s.meta.chunk_count);
        if (!need(s.group_rows.size() * 4)) { cs_set_err("truncated group rows"); return CSTRIPE_ERR_FORMAT; }
        memcpy(s.group_rows.data(), p, s.group_rows.size() * 4);
        p += s.group_rows.size() * 4;
        s.nodes.resize(head.column_count);
        for (auto &cn : s.nodes) {
            cn.resize(s.meta.chunk_count);
            for (auto &nd : cn) {
                if (!need(sizeof(nd.n))) { cs_set_err("truncated skipnode"); return CSTRIPE_ERR_FORMAT; }
                memcpy(&nd.n, p, sizeof(nd.n));
                p += sizeof(nd.n);
                if (nd.n.n_segs == 0 || nd.n.n_segs > 4096) { cs_set_err("bad n_segs"); return CSTRIPE_ERR_FORMAT; }
                nd.segs.resize(nd.n.n_segs);
                if (!need(nd.segs.size() * sizeof(csf_seg))) { cs_set_err("truncated segs"); return CSTRIPE_ERR_FORMAT; }
                memcpy(nd.segs.data(), p, nd.segs.size() * sizeof(csf_seg));
                p += nd.segs.size() * sizeof(csf_seg);
                /* split the packed decomp_len: 24-bit length + stream-shape
                 * mode (format.h); v1 files carry mode 0 = generic */
                nd.seg_modes.resize(nd.segs.size());
                for (size_t si2 = 0; si2 < nd.segs.size(); si2++) {
                    nd.seg_modes[si2] = csf_seg_mode(&nd.segs[si2]);
                    nd.segs[si2].decomp_len &= CSF_SEG_DLEN_MASK;
                }
            }
        }
    }
    return CSTRIPE_OK;
}

static int map_one(cstripe_reader *r, const char *path)
{
    int fd = open(path, O_RDONLY);
    if (fd < 0) { cs_set_err("open %s failed", path); return CSTRIPE_ERR_IO; }
    struct stat st;
    if (fstat(fd, &st) != 0 || st.st_size <= 0) { cs_set_err("stat %s failed", path); close(fd); return CSTRIPE_ERR_IO; }
    void *m = mmap(nullptr, (size_t)st.st_size, PROT_READ, MAP_PRIVATE, fd, 0);
    if (m == MAP_FAILED) { cs_set_err("mmap %s failed", path); close(fd); return CSTRIPE_ERR_IO; }
    cs_file fl;
    fl.fd = fd;
    fl.map = (const uint8_t *)m;
    fl.map_size = (size_t)st.st_size;
    r->files.push_back(fl);
    return CSTRIPE_OK;
}

/* path may be one stripe file, or a DIRECTORY of shard files (*.cs, sorted
 * by name) scanned as one table — the static shard-group-per-GPU mapping of
 * SURVEY §8e (the reference's shard placements, adaptive_executor.c) */
extern "C" cstripe_reader *cstripe_open(const char *path)
{
    auto *r = new cstripe_reader();
    struct stat st;
    if (stat(path, &st) != 0) { cs_set_err("stat %s failed", path); delete r; return nullptr; }
    if (S_ISDIR(st.st_mode)) {
        DIR *d = opendir(path);
        if (!d) { cs_set_err("opendir %s failed", path); delete r; return nullptr; }
        std::vector<std::string> names;
        struct dirent *de;
        while ((de = readdir(d)) != nullptr) {
            std::string n = de->d_name;
            if (n.size() > 3 && n.substr(n.size() - 3) == ".cs")
                names.push_back(std::string(path) + "/" + n);
        }
        closedir(d);
        std::sort(names.begin(), names.end());
        if (names.empty()) { cs_set_err("no *.cs shard files in %s", path); delete r; return nullptr; }
        for (auto &n : names)
            if (map_one(r, n.c_str()) != CSTRIPE_OK) { cstripe_close(r); return nullptr; }
    } else {
        if (map_one(r, path) != CSTRIPE_OK) { cstripe_close(r); return nullptr; }
    }
    for (uint32_t i = 0; i < r->files.size(); i++)
        if (footer_read_one(r, i) != CSTRIPE_OK) { cstripe_close(r); return nullptr; }
    return r;
}

extern "C" void cstripe_close(cstripe_reader *r)
{
    if (!r) return;
    for (auto &fl : r->files) {
        if (fl.map) munmap((void *)fl.map, fl.map_size);
        if (fl.fd >= 0) close(fl.fd);
    }
    delete r;
}

extern "C" uint64_t cstripe_row_count(const cstripe_reader *r) { return r ? r->head.total_rows : 0; }
extern "C" uint32_t cstripe_column_count(const cstripe_reader *r) { return r ? r->head.column_count : 0; }
extern "C" uint32_t cstripe_stripe_count(const cstripe_reader *r) { return r ? r->head.stripe_count : 0; }

extern "C" int cstripe_column_def(const cstripe_reader *r, uint32_t col, cstripe_coldef *out)
{
    if (!r || col >= r->cols.size() || !out) return CSTRIPE_ERR_ARG;
    memcpy(out->name, r->cols[col].name, 32);
    out->type = r->cols[col].type;
    out->scale = r->cols[col].scale;
    return CSTRIPE_OK;
}

/* ============================ pruning / scan ============================ */

/* range refutation of one predicate against a chunk's [min,max] — the
 * BuildBaseConstraint/UpdateConstraint/predicate_refuted_by combination of
 * SelectedChunkMask (columnar_reader.c:1132-1187) specialized to the
 * pushdownable family. Returns true if NO row in [min,max] can satisfy. */
static bool pred_refutes(const cstripe_pred &p, uint8_t type, int64_t min_i, int64_t max_i)
{
    if (type == CSTRIPE_F32 || type == CSTRIPE_F64) {
        double mn, mx;
        memcpy(&mn, &min_i, 8);
        memcpy(&mx, &max_i, 8);
        double c = p.fval;
        /* PG float ordering throughout (NaN high, NaN == NaN) — the
         * reference goes through predicate_refuted_by with float8 operator
         * semantics, so e.g. a chunk whose only rows above c are NaN must
         * NOT be pruned for `col > c` (its max IS NaN under this order) */
        switch (p.op) {
            case CSTRIPE_PRED_LT: return f64_cmp_pg(mn, c) >= 0;
            case CSTRIPE_PRED_LE: return f64_cmp_pg(mn, c) > 0;
            case CSTRIPE_PRED_GT: return f64_cmp_pg(mx, c) <= 0;
            case CSTRIPE_PRED_GE: return f64_cmp_pg(mx, c) < 0;
            case CSTRIPE_PRED_EQ: return f64_cmp_pg(c, mn) < 0 || f64_cmp_pg(c, mx) > 0;
            case CSTRIPE_PRED_NE: return f64_cmp_pg(mn, c) == 0 && f64_cmp_pg(mx, c) == 0;
        }
        return false;
    }
    if (type == CSTRIPE_TEXT) {
        /* TEXT min/max hold the C-collation lex key (format.h); EQ/NE are
         * order-independent, so refutation through ANY consistent order is
         * sound — range operators are never refuted (the ABI's row-level
         * text ordering is whole-slot, not collation) */
        const int64_t c = csf_text_lex_key((uint32_t)p.ival);
        switch (p.op) {
            case CSTRIPE_PRED_EQ: return c < min_i || c > max_i;
            case CSTRIPE_PRED_NE: return min_i == c && max_i == c;
            default:              return false;
        }
    }
    int64_t c = p.ival;
    switch (p.op) {
        case CSTRIPE_PRED_LT: return min_i >= c;
        case CSTRIPE_PRED_LE: return min_i > c;
        case CSTRIPE_PRED_GT: return max_i <= c;
        case CSTRIPE_PRED_GE: return max_i < c;
        case CSTRIPE_PRED_EQ: return c < min_i || c > max_i;
        case CSTRIPE_PRED_NE: return min_i == c && max_i == c;
    }
    return false;
}

extern "C" cstripe_scan *cstripe_scan_begin(cstripe_reader *r, uint64_t cols_mask,
                                            const cstripe_pred *preds, uint32_t n_preds)
{
    if (!r) { cs_set_err("scan_begin: null reader"); return nullptr; }
    auto *s = new cstripe_scan();
    s->r = r;
    s->cols_mask = cols_mask;
    if (n_preds > CSTRIPE_MAX_PREDS) { cs_set_err("too many predicates (max %d)", CSTRIPE_MAX_PREDS); delete s; return nullptr; }
    for (uint32_t i = 0; i < n_preds; i++) {
        if (preds[i].column >= r->cols.size()) { cs_set_err("pred column out of range"); delete s; return nullptr; }
        s->preds.push_back(preds[i]);
        /* normalize: every predicate belongs to exactly one OR group;
         * standalone conjuncts each get a private id */
        if (s->preds.back().or_group == 0)
            s->preds.back().or_group = 0xFFFFFF00u + i;   /* private ids */
        s->cols_mask |= 1ull << preds[i].column;   /* pred columns must be read */
    }
    /* contiguous groups (stable: original order kept within a group) */
    std::stable_sort(s->preds.begin(), s->preds.end(),
                     [](const cstripe_pred &a, const cstripe_pred &b) {
                         return a.or_group < b.or_group;
                     });
    if (s->cols_mask == 0)
        s->cols_mask = 1;   /* pure count(*): still scan one column's chunks */
    /* device pruning (SelectedChunkMask on GPU): same CNF
     * refutation evaluated one-thread-per-chunk; worthwhile when the
     * footer directory holds many thousands of chunk groups. Identical
     * semantics pinned by tests (CSTRIPE_DEVICE_PRUNE=1/0 differential). */
    {
        uint64_t total_chunks = 0;
        for (const auto &st : r->stripes) total_chunks += st.meta.chunk_count;
        const char *dp = getenv("CSTRIPE_DEVICE_PRUNE");
        const int dpv = dp ? atoi(dp) : -1;
        bool any_text = false;   /* TEXT refutation runs host-side (lex-key
                                  * EQ/NE only; kernel not taught the
                                  * transform) */
        for (const auto &p : s->preds)
            any_text |= r->cols[p.column].type == CSTRIPE_TEXT;
        std::vector<uint8_t> selmask;
        if (!s->preds.empty() && !any_text && dpv != 0 &&
            (dpv == 1 || total_chunks >= 8192) &&
            csgpu_prune(r, s->preds, selmask) == CSTRIPE_OK &&
            selmask.size() == total_chunks) {
            uint64_t ci = 0;
            for (uint32_t si = 0; si < r->stripes.size(); si++) {
                const cs_stripe_info &st = r->stripes[si];
                for (uint32_t k = 0; k < st.meta.chunk_count; k++, ci++) {
                    if (selmask[ci]) s->sel.push_back({si, k});
                    else s->chunk_groups_filtered++;
                }
            }
            return s;
        }
    }
    for (uint32_t si = 0; si < r->stripes.size(); si++) {
        const cs_stripe_info &st = r->stripes[si];
        for (uint32_t k = 0; k < st.meta.chunk_count; k++) {
            /* chunk removed iff some OR group is WHOLLY refuted — the CNF
             * equivalent of predicate_refuted_by over the reference's
             * AND/OR clause tree (SelectedChunkMask,
             * columnar_reader.c:1132-1187; OR recursion
             * columnar_customscan.c:770-829). Count a removed chunk once. */
            bool selected = true;
            size_t i = 0;
            while (i < s->preds.size() && selected) {
                size_t j = i;
                bool group_refuted = true;
                while (j < s->preds.size() &&
                       s->preds[j].or_group == s->preds[i].or_group) {
                    const cstripe_pred &p = s->preds[j];
                    const csf_skipnode &nd = st.nodes[p.column][k].n;
                    /* all-NULL chunks have no min/max and are never refuted
                     * (columnar_reader.c:1160-1166) */
                    if (!nd.has_min_max ||
                        !pred_refutes(p, r->cols[p.column].type, nd.min_i, nd.max_i))
                        group_refuted = false;
                    j++;
                }
                if (group_refuted) selected = false;
                i = j;
            }
            if (selected) s->sel.push_back({si, k});
            else s->chunk_groups_filtered++;
        }
    }
    return s;
}

extern "C" void cstripe_scan_end(cstripe_scan *s)
{
    if (!s) return;
    csgpu_release(s);
    delete s;
}

extern "C" int64_t cstripe_scan_chunk_groups_filtered(const cstripe_scan *s)
{
    return s ? s->chunk_groups_filtered : 0;
}

extern "C" int cstripe_scan_rewind(cstripe_scan *s)
{
    if (!s) return CSTRIPE_ERR_ARG;
    s->batch_pos = 0;
    return CSTRIPE_OK;
}

extern "C" int cstripe_scan_last_fused(const cstripe_scan *s) { return s ? s->last_fused : 0; }
extern "C" double cstripe_scan_last_kernel_ms(const cstripe_scan *s) { return s ? s->last_kernel_ms : 0; }
extern "C" double cstripe_scan_last_decode_kernel_ms(const cstripe_scan *s) { return s ? s->last_decode_ms : 0; }
extern "C" double cstripe_scan_last_agg_kernel_ms(const cstripe_scan *s) { return s ? s->last_agg_ms : 0; }

extern "C" int cstripe_gpu_stage(cstripe_scan *s, int device_id)
{
    if (!s) return CSTRIPE_ERR_ARG;
    return csgpu_stage(s, device_id);
}

extern "C" uint64_t cstripe_gpu_staged_bytes(const cstripe_scan *s)
{
    return s ? csgpu_staged_bytes(s) : 0;
}

extern "C" int cstripe_scan_agg(cstripe_scan *s, const cstripe_agg_spec *aggs, uint32_t n_aggs,
                                cstripe_partial *out)
{
    if (!s || !aggs || n_aggs == 0 || !out) { cs_set_err("scan_agg: bad args"); return CSTRIPE_ERR_ARG; }
    return csgpu_agg(s, aggs, n_aggs, nullptr, 0, nullptr, out);
}

extern "C" int cstripe_scan_agg_grouped(cstripe_scan *s, const cstripe_agg_spec *aggs,
                                        uint32_t n_aggs, const uint32_t *group_cols,
                                        uint32_t n_group_cols, cstripe_group_result *gr,
                                        cstripe_partial *out)
{
    if (!s || !aggs || n_aggs == 0 || !out || !gr || n_group_cols == 0 ||
        n_group_cols > CSTRIPE_MAX_GROUP_COLS) { cs_set_err("scan_agg_grouped: bad args"); return CSTRIPE_ERR_ARG; }
    return csgpu_agg(s, aggs, n_aggs, group_cols, n_group_cols, gr, out);
}

extern "C" int cstripe_read_row(cstripe_scan *s, uint64_t row_number,
                                void **col_values, uint8_t *col_nulls)
{
    if (!s || !col_values) { cs_set_err("read_row: bad args"); return CSTRIPE_ERR_ARG; }
    if (!s->preds.empty()) {
        cs_set_err("read_row: random access requires a predicate-free scan "
                   "(the reference's ColumnarReadRowByRowNumber reads with "
                   "empty clause lists)");
        return CSTRIPE_ERR_ARG;
    }
    cstripe_reader *r = s->r;
    /* locate the containing selected chunk (scan order = global row order;
     * with no predicates every chunk is selected) */
    if (s->rr_gi < 0 || row_number < s->rr_first ||
        row_number >= s->rr_first + s->rr_rows) {
        uint64_t row_base = 0;
        int64_t gi = -1;
        uint64_t hit_base = 0;
        uint64_t idx = 0;
        for (const auto &sc : s->sel) {
            const cs_stripe_info &st = r->stripes[sc.stripe];
            const uint64_t rows = st.group_rows[sc.chunk];
            if (row_number < row_base + rows) {
                gi = (int64_t)idx;
                hit_base = row_base;
                break;
            }
            row_base += rows;
            idx++;
        }
        if (gi < 0) return CSTRIPE_END;         /* no such row */
        const uint32_t n_cols = r->head.column_count;
        const uint32_t cap = r->head.chunk_row_limit;
        if (s->rr_vals.empty()) {
            s->rr_vals.resize(n_cols);
            s->rr_nulls.resize(n_cols);
            for (uint32_t c = 0; c < n_cols; c++) {
                if (!(s->cols_mask & (1ull << c))) continue;
                s->rr_vals[c].resize((size_t)cap * type_width(r->cols[c].type));
                s->rr_nulls[c].resize(cap);
            }
        }
        std::vector<void *> vp(n_cols, nullptr);
        std::vector<uint8_t *> np(n_cols, nullptr);
        for (uint32_t c = 0; c < n_cols; c++) {
            if (!(s->cols_mask & (1ull << c))) continue;
            vp[c] = s->rr_vals[c].data();
            np[c] = s->rr_nulls[c].data();
        }
        cstripe_batch b{};
        b.col_values = vp.data();
        b.col_nulls = np.data();
        int rc = csgpu_fetch_batch(s, (uint32_t)gi, &b);
        if (rc != CSTRIPE_OK) return rc;
        s->rr_gi = gi;
        s->rr_rows = b.n_rows;
        s->rr_first = hit_base;
    }
    const uint64_t off = row_number - s->rr_first;
    for (uint32_t c = 0; c < r->head.column_count; c++) {
        if (!col_values[c]) continue;
        if (!(s->cols_mask & (1ull << c))) { cs_set_err("read_row: column %u not projected", c); return CSTRIPE_ERR_ARG; }
        const uint32_t w = type_width(r->cols[c].type);
        memcpy(col_values[c], s->rr_vals[c].data() + off * w, w);
        if (col_nulls) col_nulls[c] = s->rr_nulls[c][off];
    }
    return CSTRIPE_OK;
}

extern "C" int cstripe_scan_next_batch(cstripe_scan *s, cstripe_batch *batch)
{
    if (!s || !batch) return CSTRIPE_ERR_ARG;
    return csgpu_next_batch(s, batch);
}

/* ============================ combine ============================ */

static bool agg_is_count(uint32_t k)
{
    return k == CSTRIPE_AGG_COUNT_STAR || k == CSTRIPE_AGG_COUNT_COL;
}

extern "C" int cagg_combine(const cstripe_agg_spec *aggs, uint32_t n_aggs,
                            const cstripe_partial *parts, uint32_t n_parts,
                            cstripe_partial *out)
{
    if (!aggs || !parts || !out || n_aggs == 0) { cs_set_err("combine: bad args"); return CSTRIPE_ERR_ARG; }
    for (uint32_t a = 0; a < n_aggs; a++) {
        cstripe_partial acc{};
        acc.is_null = 1;
        for (uint32_t p = 0; p < n_parts; p++) {
            const cstripe_partial &in = parts[p * n_aggs + a];
            if (in.is_null) continue;    /* strict combine skips NULL partials
                                          * (aggregate_utils.c:976-1000) */
            if (acc.is_null) { acc = in; continue; }
            switch (aggs[a].kind) {
                case CSTRIPE_AGG_COUNT_STAR:
                case CSTRIPE_AGG_COUNT_COL:
                    acc.count += in.count;
                    acc.i128_lo = acc.count;     /* keep the value mirror consistent */
                    break;
                case CSTRIPE_AGG_SUM_I64:
                case CSTRIPE_AGG_SUM_PROD_I64:
                case CSTRIPE_AGG_SUM_DISC_I64:
                case CSTRIPE_AGG_SUM_DISC_TAX_I64: {
                    __int128 x = ((__int128)acc.i128_hi << 64) | (unsigned long long)acc.i128_lo;
                    __int128 y = ((__int128)in.i128_hi << 64) | (unsigned long long)in.i128_lo;
                    x += y;
                    acc.i128_lo = (int64_t)(uint64_t)x;
                    acc.i128_hi = (int64_t)(x >> 64);
                    acc.count += in.count;
                    break;
                }
                case CSTRIPE_AGG_SUM_F64:
                    acc.f64 += in.f64;
                    acc.count += in.count;
                    break;
                case CSTRIPE_AGG_MIN_I64:
                    acc.i128_lo = std::min(acc.i128_lo, in.i128_lo);
                    acc.i128_hi = acc.i128_lo < 0 ? -1 : 0;
                    acc.count += in.count;
                    break;
                case CSTRIPE_AGG_MAX_I64:
                    acc.i128_lo = std::max(acc.i128_lo, in.i128_lo);
                    acc.i128_hi = acc.i128_lo < 0 ? -1 : 0;
                    acc.count += in.count;
                    break;
                case CSTRIPE_AGG_MIN_F64:
                    if (f64_cmp_pg(in.f64, acc.f64) < 0) acc.f64 = in.f64;
                    acc.count += in.count;
                    break;
                case CSTRIPE_AGG_MAX_F64:
                    if (f64_cmp_pg(in.f64, acc.f64) > 0) acc.f64 = in.f64;
                    acc.count += in.count;
                    break;
                default:
                    cs_set_err("combine: bad agg kind %u", aggs[a].kind);
                    return CSTRIPE_ERR_ARG;
            }
        }
        /* COUNT: NULL -> 0 (COALESCE, multi_logical_optimizer.c:1874-1884) */
        if (acc.is_null && agg_is_count(aggs[a].kind)) {
            acc.is_null = 0;
            acc.count = 0;
        }
        out[a] = acc;
    }
    return CSTRIPE_OK;
}
