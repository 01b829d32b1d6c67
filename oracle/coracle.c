/*
 * coracle.c — CPU ORACLE: a plain-C restatement of the Citus columnar read +
 * partial-aggregate path, used as (a) the parity checker for the HIP/GPU
 * path and (b) the reported CPU baseline (bench.py cpu_baseline leg).
 *
 * TEST INFRASTRUCTURE ONLY. Nothing in the product path
 * (citus_amd/libcstripe.so) calls, links or routes through this file; only
 * tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may use it.
 *
 * Function-by-function restatement of (all paths relative to /root/reference):
 *   - stripe iteration + per-row pull loop:
 *       ColumnarReadNextRow   src/backend/columnar/columnar_reader.c:322-361
 *       ReadStripeNextRow     columnar_reader.c:776-817
 *       ReadChunkGroupNextRow columnar_reader.c:868-901 (row-aligned
 *         valueArray/existsArray fill; per-row null init)
 *   - chunk-group pruning:
 *       SelectedChunkMask     columnar_reader.c:1132-1187 (min/max range
 *         refutation per qual var; all-NULL chunks never refuted; a chunk is
 *         counted filtered once)
 *   - chunk decode:
 *       DeserializeChunkData  columnar_reader.c:1583-1654
 *       DecompressBuffer      columnar_compression.c:165-270 (LZ4_decompress_safe
 *         :183, ZSTD_decompress :207 — same system libraries the reference
 *         links; one call per stored segment, segments being this format's
 *         parallel-decode extension, n_segs=1 == the reference's whole-chunk
 *         block)
 *       DeserializeBoolArray  columnar_reader.c:1506-1534 (LSB-first bits)
 *       DeserializeDatumArray columnar_reader.c:1542-1572 (packed present
 *         values walk; fixed widths here so att_align_nominal == width)
 *   - qual evaluation: ExecQual under ColumnarScanNext
 *       columnar_customscan.c:1854-1913 — SQL semantics: NULL operand fails
 *   - aggregate transitions (worker partial):
 *       multi_logical_optimizer.c:1807-1885, 2231-2275 (COUNT->count,
 *       SUM/MIN/MAX same both levels, AVG->(sum,count));
 *       NUMERIC sums exact -> __int128 fixed-point accumulators
 *   - combine step: coord_combine_agg aggregate_utils.c:820-1021 (strict:
 *       NULL partials skipped, first non-NULL initializes); COUNT NULL->0
 *       multi_logical_optimizer.c:1874-1884
 *
 * PARITY PINNING: the oracle is pinned against the reference's own golden
 * vectors (see tests/test_oracle.py):
 *   - chunk filtering counts: expected/columnar_chunk_filtering.out:132-140
 *     (i in [0,234567], chunk 10000, WHERE i>123456 -> rows 111111, chunk
 *     groups removed 12)
 *   - TPC-H Q6 revenue 243277.7858 over the 12k-row sample lineitem
 *     (expected/multi_tpch_query6.out:14-17; fixture committed under
 *     tests/golden/)
 *   - TPC-H Q1 full result set (expected/multi_tpch_query1.out)
 *
 * Build: gcc -O2 (plain C11); links system liblz4.so.1 / libzstd.so.1 via
 * hand-declared prototypes (no dev headers in the image).
 */
#include <stdint.h>
#include <stddef.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <fcntl.h>
#include <unistd.h>
#include <sys/mman.h>
#include <sys/stat.h>

/* POD shapes of the C-ABI boundary (types only; the oracle does NOT link
 * libcstripe) */
#include "../include/cstripe.h"
#include "../citus_amd/csrc/format.h"
#include "../citus_amd/csrc/pglz.h"

/* system codec prototypes (stable ABIs; reference links the same functions) */
int LZ4_decompress_safe(const char *src, char *dst, int compressedSize, int dstCapacity);
size_t ZSTD_decompress(void *dst, size_t dstCapacity, const void *src, size_t srcSize);
unsigned ZSTD_isError(size_t code);

/* ------------------------------------------------------------------ */

typedef struct onode {
    csf_skipnode n;
    csf_seg *segs;
} onode;

typedef struct ostripe {
    csf_stripe_meta meta;
    uint32_t *group_rows;
    onode *nodes;              /* [col * chunk_count + chunk] */
} ostripe;

typedef struct oracle_table {
    int fd;
    const uint8_t *map;
    size_t map_size;
    csf_footer_head head;
    csf_coldef *cols;
    ostripe *stripes;
} oracle_table;

static int o_parse_footer(oracle_table *t)
{
    if (t->map_size < CSF_HEADER_SIZE + 16) return -1;
    if (memcmp(t->map, CSF_MAGIC, 8) != 0) return -1;
    if (memcmp(t->map + t->map_size - 8, CSF_FOOT_MAGIC, 8) != 0) return -1;
    uint64_t foff;
    memcpy(&foff, t->map + t->map_size - 16, 8);
    if (foff + 16 > t->map_size) return -1;
    const uint8_t *p = t->map + foff;
    memcpy(&t->head, p, sizeof(t->head));
    p += sizeof(t->head);
    if (t->head.version != CSF_VERSION || t->head.column_count == 0) return -1;
    t->cols = calloc(t->head.column_count, sizeof(csf_coldef));
    memcpy(t->cols, p, t->head.column_count * sizeof(csf_coldef));
    p += t->head.column_count * sizeof(csf_coldef);
    t->stripes = calloc(t->head.stripe_count, sizeof(ostripe));
    for (uint32_t s = 0; s < t->head.stripe_count; s++) {
        ostripe *st = &t->stripes[s];
        memcpy(&st->meta, p, sizeof(st->meta));
        p += sizeof(st->meta);
        st->group_rows = calloc(st->meta.chunk_count, 4);
        memcpy(st->group_rows, p, st->meta.chunk_count * 4u);
        p += st->meta.chunk_count * 4u;
        st->nodes = calloc((size_t)t->head.column_count * st->meta.chunk_count, sizeof(onode));
        for (uint32_t c = 0; c < t->head.column_count; c++) {
            for (uint32_t k = 0; k < st->meta.chunk_count; k++) {
                onode *nd = &st->nodes[(size_t)c * st->meta.chunk_count + k];
                memcpy(&nd->n, p, sizeof(nd->n));
                p += sizeof(nd->n);
                nd->segs = calloc(nd->n.n_segs, sizeof(csf_seg));
                memcpy(nd->segs, p, (size_t)nd->n.n_segs * sizeof(csf_seg));
                p += (size_t)nd->n.n_segs * sizeof(csf_seg);
                /* decomp_len bits 24-31 are a stream-shape hint for the GPU
                 * (format.h); the oracle decodes every mode through the same
                 * generic LZ4/zstd call, so just mask the length */
                for (uint16_t si = 0; si < nd->n.n_segs; si++)
                    nd->segs[si].decomp_len &= CSF_SEG_DLEN_MASK;
            }
        }
    }
    return 0;
}

oracle_table *oracle_open(const char *path)
{
    int fd = open(path, O_RDONLY);
    if (fd < 0) return NULL;
    struct stat sb;
    if (fstat(fd, &sb) != 0) { close(fd); return NULL; }
    void *m = mmap(NULL, (size_t)sb.st_size, PROT_READ, MAP_PRIVATE, fd, 0);
    if (m == MAP_FAILED) { close(fd); return NULL; }
    oracle_table *t = calloc(1, sizeof(*t));
    t->fd = fd;
    t->map = m;
    t->map_size = (size_t)sb.st_size;
    if (o_parse_footer(t) != 0) {
        munmap(m, t->map_size);
        close(fd);
        free(t);
        return NULL;
    }
    return t;
}

void oracle_close(oracle_table *t)
{
    if (!t) return;
    for (uint32_t s = 0; s < t->head.stripe_count; s++) {
        ostripe *st = &t->stripes[s];
        for (uint32_t i = 0; i < t->head.column_count * st->meta.chunk_count; i++)
            free(st->nodes[i].segs);
        free(st->nodes);
        free(st->group_rows);
    }
    free(t->stripes);
    free(t->cols);
    munmap((void *)t->map, t->map_size);
    close(t->fd);
    free(t);
}

uint64_t oracle_row_count(const oracle_table *t) { return t->head.total_rows; }
uint32_t oracle_column_count(const oracle_table *t) { return t->head.column_count; }

/* DecompressBuffer restatement: per stored segment, LZ4_decompress_safe /
 * ZSTD_decompress into the exact decompressed_size (columnar_compression.c:
 * 165-270); COMPRESSION_NONE returns buffer as-is (:172-175). Returns
 * malloc'd buffer (or pointer into map for NONE; *needs_free says). */
static const uint8_t *o_decompress(const oracle_table *t, const ostripe *st,
                                   const onode *nd, int *needs_free)
{
    const uint8_t *src = t->map + st->meta.file_offset + nd->n.value_off;
    if (nd->n.comp_type == CSTRIPE_COMP_NONE) {
        *needs_free = 0;
        return src;
    }
    uint8_t *out = malloc(nd->n.decompressed_size ? nd->n.decompressed_size : 1);
    for (uint16_t i = 0; i < nd->n.n_segs; i++) {
        const csf_seg *sg = &nd->segs[i];
        if (nd->n.comp_type == CSTRIPE_COMP_LZ4) {
            int r = LZ4_decompress_safe((const char *)src + sg->comp_off,
                                        (char *)out + sg->decomp_off,
                                        (int)sg->comp_len, (int)sg->decomp_len);
            if (r != (int)sg->decomp_len) { free(out); *needs_free = 0; return NULL; }
        } else if (nd->n.comp_type == CSTRIPE_COMP_ZSTD) {
            size_t r = ZSTD_decompress(out + sg->decomp_off, sg->decomp_len,
                                       src + sg->comp_off, sg->comp_len);
            if (ZSTD_isError(r) || r != sg->decomp_len) { free(out); *needs_free = 0; return NULL; }
        } else if (nd->n.comp_type == CSTRIPE_COMP_PGLZ) {
            /* ColumnarCompressHeader (varlena + rawsize) then the pglz
             * stream (columnar_compression.c:230-262) */
            const uint8_t *pb = src + sg->comp_off;
            if (sg->comp_len < CSPGLZ_HDRSZ ||
                cspglz_varsize(pb) != sg->comp_len ||
                cspglz_rawsize(pb) != (int32_t)sg->decomp_len ||
                cspglz_decompress(pb + CSPGLZ_HDRSZ,
                                  (int32_t)(sg->comp_len - CSPGLZ_HDRSZ),
                                  out + sg->decomp_off,
                                  (int32_t)sg->decomp_len) < 0) {
                free(out); *needs_free = 0; return NULL;
            }
        } else {
            free(out); *needs_free = 0; return NULL;
        }
    }
    *needs_free = 1;
    return out;
}

/* DeserializeBoolArray (columnar_reader.c:1506-1534) */
static void o_unpack_bools(const uint8_t *buf, uint8_t *out, uint32_t n)
{
    for (uint32_t i = 0; i < n; i++)
        out[i] = (buf[i / 8] >> (i % 8)) & 1;
}

/* fetch row-aligned values of one chunk of one column.
 * values_out: rows*width bytes (null slots zeroed), exists_out: rows bytes.
 * DeserializeDatumArray walk (columnar_reader.c:1542-1572) +
 * ReadChunkGroupNextRow row-alignment (:868-901). */
int oracle_read_chunk(const oracle_table *t, uint32_t stripe, uint32_t chunk,
                      uint32_t col, void *values_out, uint8_t *exists_out)
{
    if (stripe >= t->head.stripe_count || col >= t->head.column_count) return -1;
    const ostripe *st = &t->stripes[stripe];
    if (chunk >= st->meta.chunk_count) return -1;
    const onode *nd = &st->nodes[(size_t)col * st->meta.chunk_count + chunk];
    uint32_t rows = st->group_rows[chunk];
    uint32_t width = csf_type_width(t->cols[col].type);

    o_unpack_bools(t->map + st->meta.file_offset + nd->n.exists_off, exists_out, rows);
    int needs_free = 0;
    const uint8_t *vals = o_decompress(t, st, nd, &needs_free);
    if (!vals && nd->n.decompressed_size) return -2;
    uint64_t voff = 0;
    uint8_t *dst = values_out;
    for (uint32_t i = 0; i < rows; i++) {
        if (exists_out[i]) {
            memcpy(dst + (size_t)i * width, vals + voff, width);
            voff += width;
        } else {
            memset(dst + (size_t)i * width, 0, width);
        }
    }
    if (needs_free) free((void *)vals);
    return 0;
}

/* ---------------- pruning (SelectedChunkMask) ---------------- */

/* PG float ordering (float.c float8_cmp_internal, restated): NaN sorts
 * greater than every non-NaN and equal to itself. */
static int o_f64_cmp_pg(double a, double b)
{
    if (a > b) return 1;
    if (a < b) return -1;
    if (a == b) return 0;
    {
        const int na = (a != a), nb = (b != b);
        if (na && nb) return 0;
        return na ? 1 : -1;
    }
}

static int o_pred_refutes(const cstripe_pred *p, uint8_t type, int64_t min_i, int64_t max_i)
{
    if (type == CSTRIPE_F32 || type == CSTRIPE_F64) {
        double mn, mx;
        memcpy(&mn, &min_i, 8);
        memcpy(&mx, &max_i, 8);
        double c = p->fval;
        /* PG float ordering (NaN high): matches predicate_refuted_by with
         * float8 operator semantics */
        switch (p->op) {
            case CSTRIPE_PRED_LT: return o_f64_cmp_pg(mn, c) >= 0;
            case CSTRIPE_PRED_LE: return o_f64_cmp_pg(mn, c) > 0;
            case CSTRIPE_PRED_GT: return o_f64_cmp_pg(mx, c) <= 0;
            case CSTRIPE_PRED_GE: return o_f64_cmp_pg(mx, c) < 0;
            case CSTRIPE_PRED_EQ: return o_f64_cmp_pg(c, mn) < 0 || o_f64_cmp_pg(c, mx) > 0;
            case CSTRIPE_PRED_NE: return o_f64_cmp_pg(mn, c) == 0 && o_f64_cmp_pg(mx, c) == 0;
        }
        return 0;
    }
    if (type == CSTRIPE_TEXT) {
        /* TEXT min/max hold the C-collation lex key (format.h
         * csf_text_lex_key); only order-independent EQ/NE refute */
        const int64_t c = csf_text_lex_key((uint32_t)p->ival);
        switch (p->op) {
            case CSTRIPE_PRED_EQ: return c < min_i || c > max_i;
            case CSTRIPE_PRED_NE: return min_i == c && max_i == c;
            default:              return 0;
        }
    }
    int64_t c = p->ival;
    switch (p->op) {
        case CSTRIPE_PRED_LT: return min_i >= c;
        case CSTRIPE_PRED_LE: return min_i > c;
        case CSTRIPE_PRED_GT: return max_i <= c;
        case CSTRIPE_PRED_GE: return max_i < c;
        case CSTRIPE_PRED_EQ: return c < min_i || c > max_i;
        case CSTRIPE_PRED_NE: return min_i == c && max_i == c;
    }
    return 0;
}

/* ---------------- per-row qual + agg transitions ---------------- */

typedef struct oacc {
    __int128 i128;
    double f;
    int64_t minmax_i;
    double minmax_f;
    int64_t cnt;
} oacc;

static void oacc_init(oacc *a, uint32_t kind)
{
    memset(a, 0, sizeof(*a));
    if (kind == CSTRIPE_AGG_MIN_I64) a->minmax_i = INT64_MAX;
    if (kind == CSTRIPE_AGG_MAX_I64) a->minmax_i = INT64_MIN;
    if (kind == CSTRIPE_AGG_MIN_F64) a->minmax_f = 0.0 / 0.0;   /* NaN: PG MIN identity (NaN high) */
    if (kind == CSTRIPE_AGG_MAX_F64) a->minmax_f = -1.0 / 0.0;
}

static void oacc_to_partial(const oacc *a, uint32_t kind, cstripe_partial *o)
{
    memset(o, 0, sizeof(*o));
    o->count = a->cnt;
    o->is_null = (a->cnt == 0);
    if (o->is_null && kind != CSTRIPE_AGG_COUNT_STAR && kind != CSTRIPE_AGG_COUNT_COL)
        return;   /* NULL partial: value fields zeroed */
    switch (kind) {
        case CSTRIPE_AGG_COUNT_STAR:
        case CSTRIPE_AGG_COUNT_COL:
            o->i128_lo = a->cnt;
            o->is_null = 0;
            break;
        case CSTRIPE_AGG_SUM_F64:
            o->f64 = a->f;
            break;
        case CSTRIPE_AGG_MIN_I64:
        case CSTRIPE_AGG_MAX_I64:
            o->i128_lo = a->minmax_i;
            o->i128_hi = a->minmax_i < 0 ? -1 : 0;
            break;
        case CSTRIPE_AGG_MIN_F64:
        case CSTRIPE_AGG_MAX_F64:
            o->f64 = a->minmax_f;
            break;
        default:
            o->i128_lo = (int64_t)(uint64_t)(unsigned __int128)a->i128;
            o->i128_hi = (int64_t)(a->i128 >> 64);
            break;
    }
}

/* load value of column col at row (row-aligned arrays prepared per chunk);
 * returns 0 if NULL */
typedef struct ochunkcol {
    uint8_t *values;          /* row-aligned */
    uint8_t *exists;
    uint8_t type;
    uint8_t width;
} ochunkcol;

static int o_get(const ochunkcol *cc, uint32_t row, int64_t *iv, double *fv)
{
    if (!cc->exists[row]) return 0;
    const uint8_t *p = cc->values + (size_t)row * cc->width;
    switch (cc->type) {
        case CSTRIPE_I8:  *iv = *(const int8_t *)p;  *fv = (double)*iv; break;
        case CSTRIPE_I16: { int16_t v; memcpy(&v, p, 2); *iv = v; *fv = (double)v; break; }
        case CSTRIPE_I32: { int32_t v; memcpy(&v, p, 4); *iv = v; *fv = (double)v; break; }
        case CSTRIPE_I64: { int64_t v; memcpy(&v, p, 8); *iv = v; *fv = (double)v; break; }
        case CSTRIPE_F32: { float v; memcpy(&v, p, 4); *fv = v; *iv = 0; break; }
        case CSTRIPE_TEXT: { uint32_t v; memcpy(&v, p, 4); *iv = (int64_t)v; *fv = 0; break; }
        default:          { double v; memcpy(&v, p, 8); *fv = v; *iv = 0; break; }
    }
    return 1;
}

static int o_pred_eval(const cstripe_pred *p, uint8_t is_float, int64_t iv, double fv)
{
    if (is_float) {
        const int c = o_f64_cmp_pg(fv, p->fval);   /* PG order: NaN high, NaN==NaN */
        switch (p->op) {
            case CSTRIPE_PRED_LT: return c <  0;
            case CSTRIPE_PRED_LE: return c <= 0;
            case CSTRIPE_PRED_GT: return c >  0;
            case CSTRIPE_PRED_GE: return c >= 0;
            case CSTRIPE_PRED_EQ: return c == 0;
            default:              return c != 0;
        }
    }
    switch (p->op) {
        case CSTRIPE_PRED_LT: return iv <  p->ival;
        case CSTRIPE_PRED_LE: return iv <= p->ival;
        case CSTRIPE_PRED_GT: return iv >  p->ival;
        case CSTRIPE_PRED_GE: return iv >= p->ival;
        case CSTRIPE_PRED_EQ: return iv == p->ival;
        default:              return iv != p->ival;
    }
}

/* ---- OR-group (CNF) predicate helpers: predicates sharing a nonzero
 * or_group are a disjunction; groups are ANDed (ExtractPushdownClause's
 * AND/OR recursion, columnar_customscan.c:770-829, distributed to CNF at
 * the ABI). Standalone predicates get private groups; groups made
 * contiguous by a stable sort. ---- */
static void o_norm_preds(cstripe_pred *dst, const cstripe_pred *src, uint32_t n)
{
    for (uint32_t i = 0; i < n; i++) {
        dst[i] = src[i];
        if (dst[i].or_group == 0) dst[i].or_group = 0xFFFFFF00u + i;
    }
    /* insertion sort (n <= 16), stable */
    for (uint32_t i = 1; i < n; i++) {
        cstripe_pred key = dst[i];
        int32_t j = (int32_t)i - 1;
        while (j >= 0 && dst[j].or_group > key.or_group) {
            dst[j + 1] = dst[j];
            j--;
        }
        dst[j + 1] = key;
    }
}

/* chunk survives unless some OR group is WHOLLY refuted by min/max */
static int o_chunk_selected(const oracle_table *t, const ostripe *st, uint32_t k,
                            const cstripe_pred *preds, uint32_t n_preds)
{
    uint32_t i = 0;
    while (i < n_preds) {
        uint32_t j = i;
        int group_refuted = 1;
        while (j < n_preds && preds[j].or_group == preds[i].or_group) {
            const onode *nd = &st->nodes[(size_t)preds[j].column * st->meta.chunk_count + k];
            if (!nd->n.has_min_max ||
                !o_pred_refutes(&preds[j], t->cols[preds[j].column].type,
                                nd->n.min_i, nd->n.max_i))
                group_refuted = 0;
            j++;
        }
        if (group_refuted) return 0;
        i = j;
    }
    return 1;
}

/* row filter: AND over groups of OR over members; NULL operand fails the
 * atom (SQL three-valued logic collapses to keep/drop at the top filter) */
static int o_row_pass(const oracle_table *t, const ochunkcol *cc,
                      const cstripe_pred *preds, uint32_t n_preds, uint32_t row)
{
    uint32_t i = 0;
    while (i < n_preds) {
        uint32_t j = i;
        int gv = 0;
        while (j < n_preds && preds[j].or_group == preds[i].or_group) {
            const cstripe_pred *pr = &preds[j];
            int64_t iv; double fv;
            uint8_t ty = t->cols[pr->column].type;
            if (o_get(&cc[pr->column], row, &iv, &fv))
                gv |= o_pred_eval(pr, ty == CSTRIPE_F32 || ty == CSTRIPE_F64, iv, fv);
            j++;
        }
        if (!gv) return 0;
        i = j;
    }
    return 1;
}

/* The whole-scan oracle: stripe loop -> pruning -> chunk decode -> per-row
 * qual eval -> agg transition. Mirrors the call stack of SURVEY.md §3.1.
 * group_cols: optional (n_group_cols may be 0). For grouped mode, out must
 * hold CSTRIPE_MAX_GROUPS*n_aggs partials; group keys returned in keys_out
 * (key0 | key1<<8), n_groups_out set. Group key columns must be I8 and
 * non-null. Groups ordered by first appearance then sorted by key. */
int oracle_scan_agg(oracle_table *t, uint64_t cols_mask,
                    const cstripe_pred *preds, uint32_t n_preds,
                    const cstripe_agg_spec *aggs, uint32_t n_aggs,
                    const uint32_t *group_cols, uint32_t n_group_cols,
                    cstripe_partial *out, uint32_t *keys_out,
                    uint32_t *n_groups_out, int64_t *chunk_groups_filtered)
{
    uint32_t ncols = t->head.column_count;
    uint64_t mask = cols_mask;
    for (uint32_t i = 0; i < n_preds; i++) mask |= 1ull << preds[i].column;
    cstripe_pred npreds[CSTRIPE_MAX_PREDS];
    if (n_preds > CSTRIPE_MAX_PREDS) return -5;
    o_norm_preds(npreds, preds, n_preds);
    preds = npreds;
    for (uint32_t i = 0; i < n_aggs; i++) {
        if (aggs[i].col_a >= 0) mask |= 1ull << aggs[i].col_a;
        if (aggs[i].col_b >= 0) mask |= 1ull << aggs[i].col_b;
        if (aggs[i].col_c >= 0) mask |= 1ull << aggs[i].col_c;
    }
    for (uint32_t i = 0; i < n_group_cols; i++) mask |= 1ull << group_cols[i];

    uint32_t n_groups = 0;
    uint32_t gkeys[CSTRIPE_MAX_GROUPS];
    oacc *accs = calloc((size_t)(n_group_cols ? CSTRIPE_MAX_GROUPS : 1) * n_aggs, sizeof(oacc));
    for (uint32_t a = 0; a < n_aggs; a++) oacc_init(&accs[a], aggs[a].kind);

    int64_t filtered = 0;
    ochunkcol *cc = calloc(ncols, sizeof(ochunkcol));
    uint32_t chunk_cap = t->head.chunk_row_limit;
    for (uint32_t c = 0; c < ncols; c++) {
        if (!(mask & (1ull << c))) continue;
        cc[c].type = t->cols[c].type;
        cc[c].width = (uint8_t)csf_type_width(cc[c].type);
        cc[c].values = malloc((size_t)chunk_cap * cc[c].width);
        cc[c].exists = malloc(chunk_cap);
    }

    int rc = 0;
    for (uint32_t s = 0; s < t->head.stripe_count && rc == 0; s++) {
        const ostripe *st = &t->stripes[s];
        for (uint32_t k = 0; k < st->meta.chunk_count && rc == 0; k++) {
            /* SelectedChunkMask (OR groups: see o_chunk_selected) */
            if (!o_chunk_selected(t, st, k, preds, n_preds)) { filtered++; continue; }

            uint32_t rows = st->group_rows[k];
            for (uint32_t c = 0; c < ncols; c++) {
                if (!(mask & (1ull << c))) continue;
                if (oracle_read_chunk(t, s, k, c, cc[c].values, cc[c].exists) != 0) { rc = -2; break; }
            }
            if (rc) break;

            for (uint32_t row = 0; row < rows; row++) {
                if (!o_row_pass(t, cc, preds, n_preds, row)) continue;

                oacc *grp = accs;
                if (n_group_cols) {
                    uint32_t key = 0;
                    int64_t iv; double fv;
                    for (uint32_t gcn = 0; gcn < n_group_cols; gcn++) {
                        /* NULL keys form their own group (HashAggregate
                         * groups NULLs together); 9-bit enc, bit 8 = null.
                         * TEXT: first payload byte of the varlena slot. */
                        uint32_t enc;
                        if (o_get(&cc[group_cols[gcn]], row, &iv, &fv)) {
                            uint32_t raw = (uint32_t)iv;
                            if (cc[group_cols[gcn]].type == CSTRIPE_TEXT)
                                raw >>= 8;
                            enc = raw & 0xFF;
                        } else {
                            enc = CSTRIPE_GROUP_KEY_NULL;
                        }
                        key |= enc << (9 * gcn);
                    }
                    uint32_t gi = n_groups;
                    for (uint32_t g = 0; g < n_groups; g++) if (gkeys[g] == key) { gi = g; break; }
                    if (gi == n_groups) {
                        if (n_groups >= CSTRIPE_MAX_GROUPS) { rc = -3; break; }
                        gkeys[n_groups] = key;
                        for (uint32_t a = 0; a < n_aggs; a++)
                            oacc_init(&accs[(size_t)n_groups * n_aggs + a], aggs[a].kind);
                        n_groups++;
                    }
                    grp = &accs[(size_t)gi * n_aggs];
                }

                for (uint32_t a = 0; a < n_aggs; a++) {
                    const cstripe_agg_spec *g = &aggs[a];
                    oacc *ac = &grp[a];
                    int64_t iv, ib, ic;
                    double fv, fb, fc;
                    switch (g->kind) {
                        case CSTRIPE_AGG_COUNT_STAR: ac->cnt++; break;
                        case CSTRIPE_AGG_COUNT_COL:
                            if (o_get(&cc[g->col_a], row, &iv, &fv)) ac->cnt++;
                            break;
                        case CSTRIPE_AGG_SUM_I64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv)) { ac->i128 += iv; ac->cnt++; }
                            break;
                        case CSTRIPE_AGG_SUM_F64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv)) { ac->f += fv; ac->cnt++; }
                            break;
                        case CSTRIPE_AGG_MIN_I64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv)) { if (iv < ac->minmax_i) ac->minmax_i = iv; ac->cnt++; }
                            break;
                        case CSTRIPE_AGG_MAX_I64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv)) { if (iv > ac->minmax_i) ac->minmax_i = iv; ac->cnt++; }
                            break;
                        case CSTRIPE_AGG_MIN_F64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv)) { if (o_f64_cmp_pg(fv, ac->minmax_f) < 0) ac->minmax_f = fv; ac->cnt++; }
                            break;
                        case CSTRIPE_AGG_MAX_F64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv)) { if (o_f64_cmp_pg(fv, ac->minmax_f) > 0) ac->minmax_f = fv; ac->cnt++; }
                            break;
                        case CSTRIPE_AGG_SUM_PROD_I64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv) &&
                                o_get(&cc[g->col_b], row, &ib, &fb)) {
                                ac->i128 += (__int128)iv * ib;
                                ac->cnt++;
                            }
                            break;
                        case CSTRIPE_AGG_SUM_DISC_I64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv) &&
                                o_get(&cc[g->col_b], row, &ib, &fb)) {
                                ac->i128 += (__int128)iv * (g->one - ib);
                                ac->cnt++;
                            }
                            break;
                        case CSTRIPE_AGG_SUM_DISC_TAX_I64:
                            if (o_get(&cc[g->col_a], row, &iv, &fv) &&
                                o_get(&cc[g->col_b], row, &ib, &fb) &&
                                o_get(&cc[g->col_c], row, &ic, &fc)) {
                                ac->i128 += (__int128)iv * (g->one - ib) * (g->one + ic);
                                ac->cnt++;
                            }
                            break;
                        default: rc = -4; break;
                    }
                }
            }
        }
    }

    if (rc == 0) {
        if (n_group_cols) {
            /* sort groups by key for deterministic output */
            for (uint32_t i = 0; i + 1 < n_groups; i++)
                for (uint32_t j = i + 1; j < n_groups; j++)
                    if (gkeys[j] < gkeys[i]) {
                        uint16_t tk = gkeys[i]; gkeys[i] = gkeys[j]; gkeys[j] = tk;
                        for (uint32_t a = 0; a < n_aggs; a++) {
                            oacc ta = accs[(size_t)i * n_aggs + a];
                            accs[(size_t)i * n_aggs + a] = accs[(size_t)j * n_aggs + a];
                            accs[(size_t)j * n_aggs + a] = ta;
                        }
                    }
            for (uint32_t g = 0; g < n_groups; g++) {
                if (keys_out) keys_out[g] = gkeys[g];
                for (uint32_t a = 0; a < n_aggs; a++)
                    oacc_to_partial(&accs[(size_t)g * n_aggs + a], aggs[a].kind,
                                    &out[(size_t)g * n_aggs + a]);
            }
            if (n_groups_out) *n_groups_out = n_groups;
        } else {
            for (uint32_t a = 0; a < n_aggs; a++)
                oacc_to_partial(&accs[a], aggs[a].kind, &out[a]);
            if (n_groups_out) *n_groups_out = 1;
        }
        if (chunk_groups_filtered) *chunk_groups_filtered = filtered;
    }

    for (uint32_t c = 0; c < ncols; c++) { free(cc[c].values); free(cc[c].exists); }
    free(cc);
    free(accs);
    return rc;
}

#include <pthread.h>
#include <unistd.h>

/* All-core variant of the ungrouped scan (BASELINE.md protocol step 2:
 * "single-thread, and all-core (parallel over chunk groups)"). Plain
 * pthreads over the selected chunk-group list (NOT OpenMP: the product
 * library already carries LLVM libomp, and loading gcc's libgomp into the
 * same process is a known crash). Per-thread accumulators merged with the
 * same strict-combine semantics; integer results identical to the serial
 * oracle; f64 sums differ only in addition order (within tolerance). */

typedef struct mt_ctx {
    oracle_table *t;
    const cstripe_pred *preds;
    uint32_t n_preds;
    const cstripe_agg_spec *aggs;
    uint32_t n_aggs;
    uint64_t mask;
    const uint64_t *sel;
    uint64_t nsel;
    int nthreads;
    int tid;
    oacc *accs;            /* this thread's [n_aggs] */
    int rc;
} mt_ctx;

static void *mt_worker(void *arg)
{
    mt_ctx *cx = arg;
    oracle_table *t = cx->t;
    uint32_t ncols = t->head.column_count;
    ochunkcol *cc = calloc(ncols, sizeof(ochunkcol));
    uint32_t chunk_cap = t->head.chunk_row_limit;
    for (uint32_t c = 0; c < ncols; c++) {
        if (!(cx->mask & (1ull << c))) continue;
        cc[c].type = t->cols[c].type;
        cc[c].width = (uint8_t)csf_type_width(cc[c].type);
        cc[c].values = malloc((size_t)chunk_cap * cc[c].width);
        cc[c].exists = malloc(chunk_cap);
    }
    for (uint64_t i = (uint64_t)cx->tid; i < cx->nsel; i += (uint64_t)cx->nthreads) {
        uint32_t s = (uint32_t)cx->sel[i * 2], k = (uint32_t)cx->sel[i * 2 + 1];
        const ostripe *st = &t->stripes[s];
        uint32_t rows = st->group_rows[k];
        int bad = 0;
        for (uint32_t c = 0; c < ncols && !bad; c++) {
            if (!(cx->mask & (1ull << c))) continue;
            if (oracle_read_chunk(t, s, k, c, cc[c].values, cc[c].exists) != 0) bad = 1;
        }
        if (bad) { cx->rc = -2; continue; }
        for (uint32_t row = 0; row < rows; row++) {
            int pass = 1;
            for (uint32_t p = 0; p < cx->n_preds && pass; p++) {
                const cstripe_pred *pr = &cx->preds[p];
                int64_t iv; double fv;
                uint8_t ty = t->cols[pr->column].type;
                if (!o_get(&cc[pr->column], row, &iv, &fv)) pass = 0;
                else pass = o_pred_eval(pr, ty == CSTRIPE_F32 || ty == CSTRIPE_F64, iv, fv);
            }
            if (!pass) continue;
            for (uint32_t a = 0; a < cx->n_aggs; a++) {
                const cstripe_agg_spec *g = &cx->aggs[a];
                oacc *ac = &cx->accs[a];
                int64_t iv, ib, ic; double fv, fb, fc;
                switch (g->kind) {
                    case CSTRIPE_AGG_COUNT_STAR: ac->cnt++; break;
                    case CSTRIPE_AGG_COUNT_COL:
                        if (o_get(&cc[g->col_a], row, &iv, &fv)) ac->cnt++;
                        break;
                    case CSTRIPE_AGG_SUM_I64:
                        if (o_get(&cc[g->col_a], row, &iv, &fv)) { ac->i128 += iv; ac->cnt++; }
                        break;
                    case CSTRIPE_AGG_SUM_F64:
                        if (o_get(&cc[g->col_a], row, &iv, &fv)) { ac->f += fv; ac->cnt++; }
                        break;
                    case CSTRIPE_AGG_MIN_I64:
                        if (o_get(&cc[g->col_a], row, &iv, &fv)) { if (iv < ac->minmax_i) ac->minmax_i = iv; ac->cnt++; }
                        break;
                    case CSTRIPE_AGG_MAX_I64:
                        if (o_get(&cc[g->col_a], row, &iv, &fv)) { if (iv > ac->minmax_i) ac->minmax_i = iv; ac->cnt++; }
                        break;
                    case CSTRIPE_AGG_SUM_PROD_I64:
                        if (o_get(&cc[g->col_a], row, &iv, &fv) &&
                            o_get(&cc[g->col_b], row, &ib, &fb)) { ac->i128 += (__int128)iv * ib; ac->cnt++; }
                        break;
                    case CSTRIPE_AGG_SUM_DISC_I64:
                        if (o_get(&cc[g->col_a], row, &iv, &fv) &&
                            o_get(&cc[g->col_b], row, &ib, &fb)) { ac->i128 += (__int128)iv * (g->one - ib); ac->cnt++; }
                        break;
                    case CSTRIPE_AGG_SUM_DISC_TAX_I64:
                        if (o_get(&cc[g->col_a], row, &iv, &fv) &&
                            o_get(&cc[g->col_b], row, &ib, &fb) &&
                            o_get(&cc[g->col_c], row, &ic, &fc)) { ac->i128 += (__int128)iv * (g->one - ib) * (g->one + ic); ac->cnt++; }
                        break;
                    default: cx->rc = -4; break;
                }
            }
        }
    }
    for (uint32_t c = 0; c < ncols; c++) { free(cc[c].values); free(cc[c].exists); }
    free(cc);
    return NULL;
}

int oracle_scan_agg_mt(oracle_table *t, const cstripe_pred *preds, uint32_t n_preds,
                       const cstripe_agg_spec *aggs, uint32_t n_aggs,
                       cstripe_partial *out, int *cores_out)
{
    uint64_t mask = 0;
    for (uint32_t i = 0; i < n_preds; i++) mask |= 1ull << preds[i].column;
    cstripe_pred npreds[CSTRIPE_MAX_PREDS];
    if (n_preds > CSTRIPE_MAX_PREDS) return -5;
    o_norm_preds(npreds, preds, n_preds);
    preds = npreds;
    for (uint32_t i = 0; i < n_aggs; i++) {
        if (aggs[i].col_a >= 0) mask |= 1ull << aggs[i].col_a;
        if (aggs[i].col_b >= 0) mask |= 1ull << aggs[i].col_b;
        if (aggs[i].col_c >= 0) mask |= 1ull << aggs[i].col_c;
    }

    uint64_t nsel = 0, cap = 1024;
    uint64_t *sel = malloc(cap * 16);
    for (uint32_t s = 0; s < t->head.stripe_count; s++) {
        const ostripe *st = &t->stripes[s];
        for (uint32_t k = 0; k < st->meta.chunk_count; k++) {
            if (!o_chunk_selected(t, st, k, preds, n_preds)) continue;
            if (nsel == cap) { cap *= 2; sel = realloc(sel, cap * 16); }
            sel[nsel * 2] = s;
            sel[nsel * 2 + 1] = k;
            nsel++;
        }
    }

    long ncpu = sysconf(_SC_NPROCESSORS_ONLN);
    int nthreads = ncpu > 1 ? (int)ncpu : 1;
    if (nthreads > 64) nthreads = 64;
    oacc *tacc = calloc((size_t)nthreads * n_aggs, sizeof(oacc));
    mt_ctx *ctx = calloc(nthreads, sizeof(mt_ctx));
    pthread_t *th = calloc(nthreads, sizeof(pthread_t));
    for (int i = 0; i < nthreads; i++) {
        for (uint32_t a = 0; a < n_aggs; a++)
            oacc_init(&tacc[(size_t)i * n_aggs + a], aggs[a].kind);
        ctx[i] = (mt_ctx){t, preds, n_preds, aggs, n_aggs, mask, sel, nsel,
                          nthreads, i, &tacc[(size_t)i * n_aggs], 0};
        if (i > 0) pthread_create(&th[i], NULL, mt_worker, &ctx[i]);
    }
    mt_worker(&ctx[0]);
    int rc = ctx[0].rc;
    for (int i = 1; i < nthreads; i++) {
        pthread_join(th[i], NULL);
        if (ctx[i].rc) rc = ctx[i].rc;
    }

    for (uint32_t a = 0; a < n_aggs; a++) {
        oacc m;
        oacc_init(&m, aggs[a].kind);
        for (int i = 0; i < nthreads; i++) {
            const oacc *x = &tacc[(size_t)i * n_aggs + a];
            if (x->cnt == 0) continue;
            m.i128 += x->i128;
            m.f += x->f;
            if (aggs[a].kind == CSTRIPE_AGG_MIN_I64 && x->minmax_i < m.minmax_i) m.minmax_i = x->minmax_i;
            if (aggs[a].kind == CSTRIPE_AGG_MAX_I64 && x->minmax_i > m.minmax_i) m.minmax_i = x->minmax_i;
            if (aggs[a].kind == CSTRIPE_AGG_MIN_F64 && o_f64_cmp_pg(x->minmax_f, m.minmax_f) < 0) m.minmax_f = x->minmax_f;
            if (aggs[a].kind == CSTRIPE_AGG_MAX_F64 && o_f64_cmp_pg(x->minmax_f, m.minmax_f) > 0) m.minmax_f = x->minmax_f;
            m.cnt += x->cnt;
        }
        oacc_to_partial(&m, aggs[a].kind, &out[a]);
    }
    free(tacc);
    free(ctx);
    free(th);
    free(sel);
    if (cores_out) *cores_out = nthreads;
    return rc;
}
