/*
 * cstripe.h — C-ABI drop-in boundary for the Citus columnar scan +
 * partial-aggregate hot path, MI355X-native.
 *
 * This header restates, as plain-C POD surfaces, the two reference
 * interfaces the hot path sits behind:
 *
 *  1. Table-AM scan surface:
 *     - cstripe_scan_begin()        mirrors columnar_beginscan_extended()
 *                                   (reference: src/backend/columnar/columnar_tableam.c:205-251)
 *                                   + ColumnarBeginRead (columnar_reader.c:179-239):
 *                                   projection as a column bitmask, quals as
 *                                   pushdownable `Var op const` predicates
 *                                   (columnar_customscan.c:759-952 family).
 *     - cstripe_scan_agg()          the fused whole-scan path replacing the
 *                                   per-row loop under ColumnarScanNext
 *                                   (columnar_customscan.c:1854-1894) +
 *                                   ColumnarReadNextRow (columnar_reader.c:322-361)
 *                                   + the PG Agg transition; one partial
 *                                   accumulator row out, like one worker
 *                                   partial row per shard (multi_logical_optimizer.c:1807-1885).
 *     - cstripe_scan_next_batch()   row/batch fallback access for parity tests,
 *                                   the batch analog of ColumnarReadNextRow
 *                                   (columnar_reader.c:322-361, 868-901).
 *     - cstripe_scan_chunk_groups_filtered() mirrors
 *                                   ColumnarScanChunkGroupsFiltered (columnar_tableam.h:49-60).
 *  2. Combine surface:
 *     - cagg_combine()              reproduces the coordinator merge:
 *                                   coord_combine_agg_sfunc semantics
 *                                   (distributed/utils/aggregate_utils.c:820-1021:
 *                                   strict combine skips null partials, first
 *                                   non-null initializes) and COUNT's NULL->0
 *                                   COALESCE (multi_logical_optimizer.c:1831-1885).
 *
 * Error convention: functions returning int return 0 on success, negative on
 * error; cstripe_errmsg() returns a thread-local message (reference uses
 * ereport(ERROR) longjmp; a C ABI cannot).
 * Threading: one scan = one HIP stream; caller owns batch output buffers.
 *
 * The GPU path REQUIRES a visible MI355X: any cstripe_gpu_* / cstripe_scan_agg
 * call fails loudly (CSTRIPE_ERR_NOGPU) when no HIP device is present. There
 * is no CPU compute fallback in this library; the CPU reference lives in
 * oracle/ (test infrastructure only).
 */
#ifndef CSTRIPE_H
#define CSTRIPE_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

#define CSTRIPE_ABI_VERSION 2

/* capacity limits of one scan */
#define CSTRIPE_MAX_PREDS 16
#define CSTRIPE_MAX_AGGS  12

/* ---- error codes ---- */
#define CSTRIPE_OK            0
#define CSTRIPE_ERR          -1   /* generic; see cstripe_errmsg() */
#define CSTRIPE_ERR_IO       -2
#define CSTRIPE_ERR_FORMAT   -3
#define CSTRIPE_ERR_NOGPU    -4
#define CSTRIPE_ERR_ARG      -5
#define CSTRIPE_END           1   /* scan exhausted (next_batch) */

/* ---- physical column types (fixed-width; by-value in reference terms) ----
 * Serialization follows SerializeSingleDatum (columnar_writer.c:555-585):
 * attlen-sized, att_align_nominal-aligned, packed, PRESENT VALUES ONLY
 * (nulls omitted from the value stream, DeserializeDatumArray
 * columnar_reader.c:1542-1572). For these fixed widths aligned size == width,
 * so a value stream is a dense array of the non-null values. */
typedef enum cstripe_type {
    CSTRIPE_I8  = 1,   /* also categorical char(1) codes ("char"/dictionary byte) */
    CSTRIPE_I16 = 2,
    CSTRIPE_I32 = 3,   /* also date32 (days) */
    CSTRIPE_I64 = 4,   /* also NUMERIC(15,2) as fixed-point int64 (scale in schema) */
    CSTRIPE_F32 = 5,
    CSTRIPE_F64 = 6,
    /* short varlena text (char(1)/bpchar up to 3 payload bytes): each value
     * is stored EXACTLY as the reference serializes a short-form varlena
     * datum — 1-byte header ((total_len << 1) | 1), payload, padded to the
     * 4-byte att_align_nominal boundary (SerializeSingleDatum,
     * columnar_writer.c:555-585; read back by fetch_att/att_addlength_datum,
     * columnar_reader.c:1542-1572). With payload <= 3 every slot is exactly
     * 4 bytes, so the device reads the stream as fixed-stride u32 slots.
     * The ABI passes/returns those raw 4-byte slots; predicates compare
     * whole slots (ival = the constant's slot); GROUP BY keys are the first
     * payload byte (char(1) semantics). */
    CSTRIPE_TEXT = 7,
} cstripe_type;

/* compression codec per column chunk — columnar_compression.c:62-270.
 * Values match the reference's CompressionType enum
 * (columnar_compression.h: NONE=0, PG_LZ=1, LZ4=2, ZSTD=3). */
typedef enum cstripe_compression {
    CSTRIPE_COMP_NONE = 0,
    CSTRIPE_COMP_PGLZ = 1,   /* not produced by this writer; recognized in metadata */
    CSTRIPE_COMP_LZ4  = 2,
    CSTRIPE_COMP_ZSTD = 3,
} cstripe_compression;

/* ---- schema / options ---- */
typedef struct cstripe_coldef {
    char        name[32];
    uint8_t     type;      /* cstripe_type */
    uint8_t     scale;     /* decimal scale for fixed-point int64/int32 (display only) */
    uint8_t     _pad[6];
} cstripe_coldef;

/* knobs mirror columnar GUCs (columnar.c:30-44, :68-120):
 * stripe_row_limit=150000, chunk_group_row_limit=10000, compression, level=3 */
typedef struct cstripe_options {
    uint64_t    stripe_row_limit;      /* default 150000 */
    uint32_t    chunk_group_row_limit; /* default 10000  */
    uint8_t     compression;           /* cstripe_compression, default LZ4 */
    int8_t      compression_level;     /* default 3 (zstd) */
    uint16_t    lz4_seg_target_kb;     /* decompressed KiB per independently
                                        * decodable LZ4 segment (coarse knob;
                                        * one segment == plain whole-chunk
                                        * block, exactly the reference layout) */
    uint32_t    lz4_seg_target_bytes;  /* fine knob, overrides kb when nonzero;
                                        * default 256 B = one GPU lane per
                                        * segment (lane-parallel decode) */
    uint8_t     lz4_min_match;         /* shortest match the writer's LZ4
                                        * encoder emits (>=4; 0 => default).
                                        * Larger = literal-heavier standard-LZ4
                                        * streams that decode faster on GPU at
                                        * a small compression-ratio cost */
    uint8_t     canonical;             /* 1 (default): width-8 chunks whose
                                        * values share their high bytes are
                                        * written as a CANONICAL periodic LZ4
                                        * parse — still a standard block that
                                        * LZ4_decompress_safe decodes — whose
                                        * value positions are closed-form, so
                                        * GPU kernels read values straight
                                        * from the compressed stream with no
                                        * sequence parsing. 0: greedy parse
                                        * everywhere (round-1 behaviour). */
    uint8_t     _pad2[6];
} cstripe_options;

void cstripe_default_options(cstripe_options *opts);

/* ---- predicates ----
 * The family the reference pushes down: `Var op pseudo-const` atoms combined
 * with AND and OR (ExtractPushdownClause, columnar_customscan.c:712-952,
 * OR recursion :770-829). The ABI carries the filter in CNF: predicates
 * sharing a nonzero or_group form one DISJUNCTION; the groups (and all
 * or_group==0 predicates, each its own group) are ANDed. Any AND/OR tree of
 * atoms distributes into this form — e.g. the reference test's
 * `(a>1000 AND a<10000) OR (a>20000 AND a<50000)` becomes the four groups
 * (a>1000 | a>20000)(a>1000 | a<50000)(a<10000 | a>20000)(a<10000 | a<50000)
 * — and chunk refutation over the CNF is equivalent to the reference's
 * predicate_refuted_by on the original tree (a group prunes a chunk only if
 * EVERY member refutes its [min,max]). Trees that exceed CSTRIPE_MAX_PREDS
 * after distribution are not pushdownable through this ABI, mirroring the
 * reference leaving such clauses to ExecQual. Values are given in the
 * column's physical representation (i64 for integer/fixed-point/date
 * columns, f64 for floats). NULL operands fail the atom (SQL semantics at
 * the top-level filter). */
typedef enum cstripe_predop {
    CSTRIPE_PRED_LT = 0, CSTRIPE_PRED_LE, CSTRIPE_PRED_GT, CSTRIPE_PRED_GE,
    CSTRIPE_PRED_EQ, CSTRIPE_PRED_NE,
} cstripe_predop;

typedef struct cstripe_pred {
    uint32_t    column;    /* 0-indexed (reference uses 1-indexed attno) */
    uint32_t    op;        /* cstripe_predop */
    int64_t     ival;      /* comparison constant for integer columns */
    double      fval;      /* comparison constant for float columns */
    uint32_t    or_group;  /* 0 = standalone conjunct; same nonzero id = OR'd */
    uint32_t    _pad;
} cstripe_pred;

/* ---- aggregates ----
 * The worker-partial shapes of the built-ins on the hot path
 * (multi_logical_optimizer.c:1807-1885, :2231-2275, :3279-3308):
 * COUNT -> worker count (int64); SUM/MIN/MAX -> same agg both levels;
 * AVG -> worker (sum, count). Fixed-point NUMERIC sums use int128
 * accumulators (exact). The product forms cover TPC-H Q6/Q1 expressions. */
typedef enum cstripe_aggkind {
    CSTRIPE_AGG_COUNT_STAR = 0,  /* rows passing filter */
    CSTRIPE_AGG_COUNT_COL,       /* non-null values of col a */
    CSTRIPE_AGG_SUM_I64,         /* sum(a)                  -> i128 */
    CSTRIPE_AGG_SUM_F64,         /* sum(a)                  -> f64  */
    CSTRIPE_AGG_MIN_I64, CSTRIPE_AGG_MAX_I64,
    CSTRIPE_AGG_MIN_F64, CSTRIPE_AGG_MAX_F64,
    CSTRIPE_AGG_SUM_PROD_I64,    /* sum(a*b)                -> i128 (Q6 revenue, scale sa+sb) */
    CSTRIPE_AGG_SUM_DISC_I64,    /* sum(a*(ONE-b))          -> i128 (Q1 disc_price) */
    CSTRIPE_AGG_SUM_DISC_TAX_I64,/* sum(a*(ONE-b)*(ONE+c))  -> i128 (Q1 charge) */
} cstripe_aggkind;

typedef struct cstripe_agg_spec {
    uint32_t    kind;            /* cstripe_aggkind */
    int32_t     col_a;           /* -1 for COUNT_STAR */
    int32_t     col_b;           /* product forms */
    int32_t     col_c;           /* disc_tax form */
    int64_t     one;             /* fixed-point ONE for disc/tax forms (100 at scale 2) */
} cstripe_agg_spec;

/* One partial accumulator — the wire shape that replaces a worker partial-agg
 * row (aggregate_utils.c:706-743 serializes transition state per group; ours
 * is the POD equivalent). is_null mirrors a strict transition that saw no
 * rows (aggregate_utils.c:976-1000). */
typedef struct cstripe_partial {
    int64_t     i128_lo;         /* int128 sums: low 64 (two's complement) */
    int64_t     i128_hi;         /*              high 64 */
    double      f64;             /* float sums / min / max */
    int64_t     count;           /* COUNT kinds; also rows contributing */
    uint8_t     is_null;         /* 1 = no rows contributed (strict agg NULL) */
    uint8_t     _pad[7];
} cstripe_partial;

/* ---- group-by (Q1 shape): up to 2 categorical u8/i8 key columns ----
 * NULL keys form their own group, like the reference's HashAggregate
 * (grouping treats NULLs as equal). Key encoding: 9 bits per key column —
 * enc = (value & 0xFF) | (is_null << 8) — packed enc0 | enc1 << 9. */
#define CSTRIPE_MAX_GROUP_COLS 2
#define CSTRIPE_MAX_GROUPS     64
#define CSTRIPE_GROUP_KEY_NULL 0x100u

typedef struct cstripe_group_result {
    uint32_t    n_groups;
    uint32_t    keys[CSTRIPE_MAX_GROUPS];        /* enc0 | enc1<<9 */
    /* partials[g*n_aggs + a] for group g, agg a — caller-provided, size
     * CSTRIPE_MAX_GROUPS * n_aggs */
} cstripe_group_result;

/* ---- batch output (parity/fallback access) ----
 * Caller provides per-column destination arrays (physical width each) and a
 * nulls byte array; the batch is one chunk group (<= chunk_group_row_limit
 * rows), values row-aligned (null slots zero-filled), mirroring
 * ReadChunkGroupNextRow's columnValues/columnNulls contract
 * (columnar_reader.c:868-901) but batched. */
typedef struct cstripe_batch {
    uint32_t    n_rows;          /* out: rows in this batch */
    uint64_t    first_row_number;/* out */
    void      **col_values;      /* in: [n_cols] ptrs, each >= chunk_row_limit*width; unprojected may be NULL */
    uint8_t   **col_nulls;       /* in: [n_cols] ptrs, each >= chunk_row_limit; 1=null */
} cstripe_batch;

/* ---- opaque handles ---- */
typedef struct cstripe_writer cstripe_writer;
typedef struct cstripe_reader cstripe_reader;
typedef struct cstripe_scan   cstripe_scan;

/* =================== writer (format producer; host C) ===================
 * Produces the exact on-disk layout of FlushStripe (columnar_writer.c:391-516):
 * per stripe, per column: all exists streams (bit-packed, SerializeBoolArray
 * :523-545), then all value streams (per-chunk, compressed per
 * SerializeChunkData :592-654 / CompressBuffer, kept raw when incompressible),
 * offsets recorded in skip nodes; skip nodes + chunk-group row counts go to a
 * flat footer directory replacing the columnar.stripe/chunk_group/chunk
 * catalogs (columnar_metadata.c:604-832). */
cstripe_writer *cstripe_write_begin(const char *path, const cstripe_coldef *cols,
                                    uint32_t n_cols, const cstripe_options *opts);
/* column-batch append: n_rows rows; values[c] = array of column c's physical
 * type, row-aligned; nulls[c] may be NULL (all present) else bytes 1=null. */
int cstripe_write_rows(cstripe_writer *w, uint64_t n_rows,
                       const void *const *values, const uint8_t *const *nulls);
int cstripe_write_end(cstripe_writer *w);     /* flush + footer + close; frees w */
/* device-side append: values[c] are DEVICE pointers to
 * HBM-resident column arrays (query results / ETL). Full chunks are
 * compressed ON the GPU (canonical parses where the data fits; see
 * format.h) so only compressed bytes cross PCIe; other shapes copy back
 * and take the host compressors. Dense rows only — NULL-bearing batches
 * go through cstripe_write_rows. Requires a visible MI355X. */
int cstripe_write_rows_device(cstripe_writer *w, uint64_t n_rows,
                              const void *const *dev_values);
void cstripe_write_abort(cstripe_writer *w);

/* =================== reader / scan =================== */
cstripe_reader *cstripe_open(const char *path);          /* mmap + parse footer */
void cstripe_close(cstripe_reader *r);
uint64_t cstripe_row_count(const cstripe_reader *r);
uint32_t cstripe_column_count(const cstripe_reader *r);
uint32_t cstripe_stripe_count(const cstripe_reader *r);
int cstripe_column_def(const cstripe_reader *r, uint32_t col, cstripe_coldef *out);

/* scan_begin: projection as a bitmask over columns (bit c = column c needed),
 * preds implicitly ANDed. Performs chunk-group min/max pruning host-side
 * (SelectedChunkMask semantics, columnar_reader.c:1132-1187: a chunk survives
 * unless some predicate refutes its [min,max]; chunks with no min/max always
 * survive). */
cstripe_scan *cstripe_scan_begin(cstripe_reader *r, uint64_t cols_mask,
                                 const cstripe_pred *preds, uint32_t n_preds);
void cstripe_scan_end(cstripe_scan *s);
int64_t cstripe_scan_chunk_groups_filtered(const cstripe_scan *s);

/* GPU staging: hipMalloc + HtoD copy of the selected chunks' compressed value
 * streams and exists streams for projected columns; builds device chunk
 * descriptors. Required before cstripe_scan_agg / GPU next_batch.
 * device_id < 0 => current device. */
int cstripe_gpu_stage(cstripe_scan *s, int device_id);
/* bytes staged to HBM (compressed value streams + exists bitmaps) */
uint64_t cstripe_gpu_staged_bytes(const cstripe_scan *s);

/* The fused hot path: decode(LZ4) -> filter -> partial aggregate, on device.
 * Writes one partial per agg spec. Re-invocable (re-runs over staged data;
 * rescan semantics of columnar_rescan). Plain aggregate (no grouping). */
int cstripe_scan_agg(cstripe_scan *s, const cstripe_agg_spec *aggs, uint32_t n_aggs,
                     cstripe_partial *partials_out);

/* GROUP BY variant (Q1 shape): group_cols = up to 2 columns of type I8.
 * partials_out sized CSTRIPE_MAX_GROUPS * n_aggs; result keys/groups in gr. */
int cstripe_scan_agg_grouped(cstripe_scan *s, const cstripe_agg_spec *aggs,
                             uint32_t n_aggs, const uint32_t *group_cols,
                             uint32_t n_group_cols, cstripe_group_result *gr,
                             cstripe_partial *partials_out);

/* Random-access read — ColumnarReadRowByRowNumber
 * (columnar_reader.c:386-441): reads the row with the given row number
 * (0-based scan-order position over the file/shard directory) into the
 * caller's per-column buffers: col_values[c] points at one value slot of
 * column c's physical width (unprojected columns may be NULL), col_nulls[c]
 * gets 0/1. Returns CSTRIPE_OK, CSTRIPE_END when no such row exists (the
 * reference returns false), or an error. Requires a scan begun WITHOUT
 * predicates — the reference's random-access path passes empty clause
 * lists the same way (:423-424). The containing chunk is device-decoded
 * and cached across consecutive calls, like the reference keeping the
 * current stripe open. */
int cstripe_read_row(cstripe_scan *s, uint64_t row_number,
                     void **col_values, uint8_t *col_nulls);

/* Batch access (parity path): returns CSTRIPE_OK with a filled batch, or
 * CSTRIPE_END when exhausted. GPU-decodes chunk by chunk and copies back;
 * residual (non-pruned-chunk) predicate filtering is NOT applied — caller
 * sees all rows of surviving chunks, as ColumnarReadNextRow does before
 * ExecQual. */
int cstripe_scan_next_batch(cstripe_scan *s, cstripe_batch *batch);
int cstripe_scan_rewind(cstripe_scan *s);

/* 1 when the last cstripe_scan_agg ran the fused decode+filter+aggregate
 * kernel (no scratch round trip), 0 for the two-kernel path */
int cstripe_scan_last_fused(const cstripe_scan *s);

/* per-call timing of the last cstripe_scan_agg: kernel time (hipEvents, on the
 * scan's stream) and wall time inside the call, milliseconds. */
double cstripe_scan_last_kernel_ms(const cstripe_scan *s);
double cstripe_scan_last_decode_kernel_ms(const cstripe_scan *s);
double cstripe_scan_last_agg_kernel_ms(const cstripe_scan *s);

/* =================== combine surface =================== */
/* Merge per-shard/per-GPU partials: out = combine(parts[0..n)). Reproduces
 * coord_combine_agg semantics: strict combine over non-null partials
 * (aggregate_utils.c:976-1000); COUNT kinds get NULL->0 (COALESCE,
 * multi_logical_optimizer.c:1831-1885), so a COUNT out is never null. */
int cagg_combine(const cstripe_agg_spec *aggs, uint32_t n_aggs,
                 const cstripe_partial *parts, uint32_t n_parts,
                 cstripe_partial *out);

/* =================== RCCL combine (device collective) ===================
 * The north star replaces the coordinator merge with a collective over
 * xGMI. The data-path hop is ncclAllGather of each rank's partial block
 * (RCCL over xGMI), followed by the same strict cagg_combine merge locally
 * on every rank: int128 carries cannot ride a sum collective, and gathering
 * is byte-faithful to the coordinator receiving one partial row per shard
 * (aggregate_utils.c:820-1021; adaptive_executor.c:775-882). Host code stays
 * C; the launcher only has to distribute the 128-byte unique id (the same
 * bootstrap contract as ncclCommInitRank). */
typedef struct cagg_comm cagg_comm;
#define CAGG_UNIQUE_ID_BYTES 128

int cagg_comm_unique_id(uint8_t id[CAGG_UNIQUE_ID_BYTES]);   /* rank 0 makes */
int cagg_comm_init(cagg_comm **out, int n_ranks, int rank,
                   const uint8_t id[CAGG_UNIQUE_ID_BYTES], int device);
void cagg_comm_destroy(cagg_comm *c);
int cagg_comm_rank(const cagg_comm *c);
int cagg_comm_size(const cagg_comm *c);

/* all-gather equal-size byte blocks from every rank (device-staged
 * internally; dst holds n_ranks * bytes, rank-major) */
int cagg_allgather(cagg_comm *c, const void *src, uint64_t bytes, void *dst);

/* the combine surface over RCCL: gather every rank's partial block, then
 * cagg_combine locally; every rank returns the final merged row */
int cagg_combine_rccl(cagg_comm *c, const cstripe_agg_spec *aggs,
                      uint32_t n_aggs, const cstripe_partial *local,
                      cstripe_partial *out);

/* =================== misc =================== */
const char *cstripe_errmsg(void);
int cstripe_gpu_available(void);     /* 1 if a HIP device is visible */
uint32_t cstripe_abi_version(void);

#ifdef __cplusplus
}
#endif
#endif /* CSTRIPE_H */
