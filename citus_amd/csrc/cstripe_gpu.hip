/*
 * cstripe_gpu.hip — MI355X (gfx950/CDNA4) device path of the columnar scan:
 * HtoD staging of compressed chunk streams, LZ4 segment decode kernel, and
 * the fused predicate-filter + partial-aggregate kernels.
 *
 * This replaces, with HIP kernels, the CPU loops under ColumnarReadNextRow:
 *   - DecompressBuffer/LZ4_decompress_safe per chunk
 *     (columnar_compression.c:165-198)           -> lz4_decode_kernel
 *   - DeserializeBoolArray (columnar_reader.c:1506-1534): exists bitmaps are
 *     NOT unpacked — kernels consume the packed bits directly (rank tables
 *     precomputed host-side at stage)
 *   - DeserializeDatumArray (columnar_reader.c:1542-1572): fixed-width value
 *     streams are read as coalesced typed loads (layout guaranteed by
 *     SerializeSingleDatum, columnar_writer.c:555-585)
 *   - per-row qual eval (ExecQual under columnar_customscan.c:1907-1913)
 *     -> predicate evaluation in-register, per 64-lane wavefront
 *   - PG Agg transition + worker_partial_agg (aggregate_utils.c:501-607)
 *     -> per-thread accumulate, wave shuffle reduce, LDS cross-wave reduce,
 *        per-block partial, device-wide final reduce (no global atomics)
 *
 * All scan/filter/reduce — HBM-bandwidth bound; no MFMA (north star).
 * Wavefront = 64 throughout; __ballot is 64-bit.
 */
#include "internal.h"
#include "compress.h"
#include "pglz.h"
#include "zstd_r.h"
#include "lz4_enc.h"

#include <hip/hip_runtime.h>
#include <cstring>
#include <cstdio>
#include <vector>
#include <algorithm>

#define WAVE 64
#define AGG_BLOCK 256
#define TILE_ROWS 4096
#define MAX_PREDS CSTRIPE_MAX_PREDS
#define MAX_AGGS 12
#define MAX_PROJ 16

/* ---------------- device-side descriptors ---------------- */

struct SegDesc {
    uint64_t src_off;       /* into d_data (compressed) */
    uint64_t dst_off;       /* into d_scratch (decompressed) */
    uint32_t comp_len;
    uint32_t decomp_len;
};

struct ColLoc {
    uint64_t val_off;       /* into d_scratch if (flags&1) else d_data */
    uint64_t exists_off;    /* into d_data (packed bitmap, 8B-padded) */
    uint32_t rank_off;      /* u32 word-rank table index into d_rank */
    uint16_t flags;         /* 1 = values in scratch; 2 = dense (no nulls);
                             * 4 = canonical stream (closed-form access) */
    uint8_t  type;          /* cstripe_type */
    uint8_t  width;
    /* canonical-stream chunks (flags & 4): value bytes are read closed-form
     * straight from the staged COMPRESSED stream at val_off (format.h) */
    int64_t  hval;          /* P: shared high bytes, already in place;
                             * CONST: the value; LIT: literal-run header len */
    uint8_t  mode;          /* CSF_SEGMODE_* */
    uint8_t  L;             /* P(L): varying low bytes per value */
    uint8_t  pad[6];
};

struct GroupDesc {
    uint32_t row_count;
    uint32_t colbase;       /* index into d_colloc: colbase + proj_idx */
};

struct PredD {
    uint16_t proj;
    uint8_t  op;
    uint8_t  is_float;
    uint8_t  gend;       /* 1 = last member of its OR group (groups are
                          * contiguous; pass &= OR of the group's members) */
    uint8_t  pad[7];
    int64_t  ival;
    double   fval;
};

struct AggD {
    uint8_t kind;
    uint8_t proj_a, proj_b, proj_c;
    int64_t one;
};

struct AggParams {
    uint32_t n_preds, n_aggs, n_proj, tiles_per_group, n_groups;
    uint32_t pad[3];
    PredD preds[MAX_PREDS];
    AggD aggs[MAX_AGGS];
};

struct FusedTile;
struct FusedTileG;

/* per-block / final accumulator cell (32 B) */
struct AccCell {
    int64_t lo;             /* i128 low  / i64 min-max */
    int64_t hi;             /* i128 high */
    double  f;              /* f64 sum / min / max */
    int64_t cnt;            /* contributing rows */
};

struct cs_gpu_state {
    int device = -1;
    hipStream_t stream = nullptr;
    uint8_t *d_data = nullptr;       /* staged compressed values + exists */
    uint8_t *d_scratch = nullptr;    /* decompressed value streams */
    uint32_t *d_rank = nullptr;
    SegDesc *d_segs = nullptr;
    GroupDesc *d_groups = nullptr;
    ColLoc *d_colloc = nullptr;
    AccCell *d_block = nullptr;
    AccCell *d_final = nullptr;
    AccCell *d_final2 = nullptr;     /* wide-pass outputs of the final reduce */
    uint32_t *d_gkeys = nullptr;     /* grouped: per-block key tables */
    AccCell *d_gcells = nullptr;
    uint32_t *d_gfkeys = nullptr;    /* grouped: final merged groups */
    AccCell *d_gfcells = nullptr;
    uint32_t *d_gn = nullptr;
    uint8_t *d_tmp = nullptr;        /* next_batch: canonical-chunk decode buf */
    uint64_t tmp_bytes = 0;
    bool batch_decoded = false;      /* scratch holds decoded streams */
    SegDesc *d_zsegs = nullptr;      /* restricted-zstd segments (zstd_r.h) */
    uint32_t n_zsegs = 0;
    uint32_t max_zseg_comp = 0;      /* largest zstd frame (LDS-variant gate) */
    uint32_t max_zseg_dlen = 0;      /* largest zstd frame decompressed */
    zr_dtables *d_zrtab = nullptr;   /* predefined FSE decode tables */
    uint32_t *d_segstart = nullptr;  /* per (group, proj): first d_segs index */
    uint64_t greedy256_mask = 0;     /* proj cols stored as uniform 256B lz4 */
    std::vector<uint8_t> col_scratch;/* per proj: any chunk lands in scratch */
    int *d_error = nullptr;
    hipEvent_t ev0 = nullptr, ev1 = nullptr, ev2 = nullptr;

    uint64_t data_bytes = 0;
    uint64_t scratch_bytes = 0;
    uint32_t n_segs = 0;
    uint32_t max_seg_comp = 0;
    uint32_t max_seg_dlen = 0;
    bool segs_16aligned = true;
    bool fusable = true;             /* all proj cols dense i64, uniform 256B lz4 segs */
    bool fusable_mixed = true;       /* widths in {1,8}; enables fused grouped */
    bool all_dense = true;           /* no NULLs anywhere -> pair_agg_kernel */
    bool all_canonP = true;          /* every col canonical P/CONST -> lds kernel */
    struct FusedTile *d_tiles = nullptr;
    uint32_t n_tiles = 0;
    struct FusedTileG *d_tiles2 = nullptr;
    uint32_t n_tiles2 = 0;
    uint16_t flane_base[8] = {0};
    uint8_t fwidth[8] = {0};
    uint16_t flanes_total = 0;
    uint32_t n_groups = 0;
    uint32_t n_proj = 0;
    uint32_t max_blocks = 0;
    int proj_of_col[64];             /* column index -> projected slot (-1) */
    uint8_t proj_type[MAX_PROJ];
    /* host-side mirror for next_batch (exists read from mmap) */
    std::vector<uint64_t> scratch_off;  /* per (sel,proj): decomp offset */
    std::vector<ColLoc> colloc_host;
};

/* ---------------- error helper ---------------- */

#define HIP_TRY(x) do { hipError_t _e = (x); if (_e != hipSuccess) { \
    cs_set_err("%s failed: %s", #x, hipGetErrorString(_e)); return CSTRIPE_ERR; } } while (0)

extern "C" int cstripe_gpu_available(void)
{
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n > 0 ? 1 : 0;
}

/* =====================================================================
 * LZ4 block decode — one 64-lane wavefront per independently decodable
 * segment. All lanes parse the sequence header redundantly (same-address
 * loads broadcast); literal and match copies are lane-parallel. Overlapping
 * matches (offset < length) resolve every byte to the pre-existing pattern
 * window via modulo, so one parallel pass is race-free.
 * Implements the LZ4 block format consumed by LZ4_decompress_safe
 * (columnar_compression.c:183); any valid block decodes, segmented or not.
 *
 * Two variants:
 *  - lz4_decode_lds_kernel: compressed segment is first staged into LDS with
 *    coalesced u32 loads and the OUTPUT is decoded in LDS too, so the
 *    byte-serial parse and all match copies run at LDS latency instead of
 *    chained dependent HBM round trips; the finished segment is flushed to
 *    global with coalesced 16 B stores. Needs in+out <= 64 KiB of LDS —
 *    the writer's default 8 KiB segments give ~9 resident waves/CU.
 *  - lz4_decode_kernel: the global-memory fallback for oversized segments
 *    (e.g. single-block whole-chunk compatibility mode).
 * ===================================================================== */

template <bool IN_LDS>
__device__ inline void lz4_body(const uint8_t *__restrict__ src, uint32_t slen,
                                uint8_t *__restrict__ dst, uint32_t dlen,
                                uint32_t lane, int *__restrict__ err)
{
    uint32_t ip = 0, op = 0;
    while (ip < slen) {
        const uint32_t token = src[ip++];
        /* literals */
        uint32_t litlen = token >> 4;
        if (litlen == 15) {
            uint32_t b;
            do { b = src[ip++]; litlen += b; } while (b == 255 && ip < slen);
        }
        if (ip + litlen > slen || op + litlen > dlen) { if (lane == 0) atomicOr(err, 1); return; }
        for (uint32_t j = lane; j < litlen; j += WAVE) dst[op + j] = src[ip + j];
        ip += litlen;
        op += litlen;
        if (ip >= slen) break;           /* last sequence: literals only */

        /* match */
        if (ip + 2 > slen) { if (lane == 0) atomicOr(err, 1); return; }
        uint32_t offset = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8);
        ip += 2;
        uint32_t mlen = token & 15;
        if (mlen == 15) {
            uint32_t b;
            do { b = src[ip++]; mlen += b; } while (b == 255 && ip < slen);
        }
        mlen += 4;
        if (offset == 0 || offset > op || op + mlen > dlen) { if (lane == 0) atomicOr(err, 1); return; }
        const uint32_t mstart = op - offset;
        if (offset >= mlen) {
            for (uint32_t j = lane; j < mlen; j += WAVE) dst[op + j] = dst[mstart + j];
        } else {
            /* every byte's ultimate source lies in the pre-existing pattern
             * window [op-offset, op) — single race-free parallel pass */
            for (uint32_t j = lane; j < mlen; j += WAVE) dst[op + j] = dst[mstart + (j % offset)];
        }
        op += mlen;
    }
    if (op != dlen) { if (lane == 0) atomicOr(err, 2); }
}

__global__ __launch_bounds__(WAVE) void lz4_decode_kernel(
    const uint8_t *__restrict__ data, uint8_t *__restrict__ scratch,
    const SegDesc *__restrict__ segs, int *__restrict__ err)
{
    const SegDesc s = segs[blockIdx.x];
    lz4_body<false>(data + s.src_off, s.comp_len, scratch + s.dst_off,
                    s.decomp_len, threadIdx.x, err);
}

/* =====================================================================
 * Lane-parallel LZ4 decode — ONE LANE per micro-segment (writer default
 * 512 B decompressed). Why: this data compresses to ~one sequence per
 * 8-byte value (measured ~8 B/sequence), so a wave-cooperative decoder is
 * serialized on the per-sequence parse chain and moves ~8 B per chain step.
 * Here every wave advances 64 independent segment streams per step:
 *  - input: per-lane 16-byte register window over the compressed stream,
 *    refilled with aligned u64 loads (one dependent global load per 8
 *    compressed bytes instead of per byte)
 *  - output: per-lane LDS region (stride 528 = 16-byte aligned, 4-bank
 *    skew per lane so equal-progress lanes hit different banks); match
 *    copies are plain per-lane sequential byte moves (memmove-forward
 *    semantics handles overlap for free)
 *  - finish: block-cooperative 16 B coalesced flush LDS -> scratch
 * ===================================================================== */

struct ByteStream {
    const uint8_t *base;    /* 8B-aligned */
    uint64_t w0, w1, w2, w3; /* 32-byte window; ~24 B of refill lookahead */
    uint32_t woff;          /* byte offset of w0 within base */
    uint32_t pos;           /* current byte position (>= initial shift) */
};

__device__ inline void bs_init(ByteStream &b, const uint8_t *data, uint64_t off)
{
    uint64_t a = off & ~15ull;           /* 16B-aligned base */
    b.base = data + a;
    b.pos = (uint32_t)(off - a);
    b.woff = 0;
    b.w0 = *(const uint64_t *)(b.base);
    b.w1 = *(const uint64_t *)(b.base + 8);
    b.w2 = *(const uint64_t *)(b.base + 16);
    b.w3 = *(const uint64_t *)(b.base + 24);
}

__device__ inline void bs_norm(ByteStream &b)
{
    while (b.pos - b.woff >= 8) {
        b.w0 = b.w1;
        b.w1 = b.w2;
        b.w2 = b.w3;
        b.woff += 8;
        b.w3 = *(const uint64_t *)(b.base + b.woff + 24);
    }
}

__device__ inline uint8_t bs_get(ByteStream &b)
{
    bs_norm(b);
    uint32_t rel = b.pos - b.woff;
    b.pos++;
    return (uint8_t)(b.w0 >> (8 * rel));
}

/* next 8 bytes at pos as a little-endian u64 (rel in [0,8) after bs_norm) */
__device__ inline uint64_t bs_peek64(ByteStream &b)
{
    bs_norm(b);
    uint32_t s = 8 * (b.pos - b.woff);
    return s ? ((b.w0 >> s) | (b.w1 << (64 - s))) : b.w0;
}

/* decode ONE segment, per-lane serial, output to this lane's LDS region.
 * Shared by the standalone decode kernel and the fused decode+agg kernel. */
__device__ inline void lz4_lane_decode(const uint8_t *__restrict__ data,
                                       const SegDesc &s, uint8_t *__restrict__ sout,
                                       int *__restrict__ err)
{
    {
        ByteStream bs;
        bs_init(bs, data, s.src_off);
        const uint32_t send = bs.pos + s.comp_len;   /* end position */
        const uint32_t dlen = s.decomp_len;
        uint32_t op = 0;
        bool bad = false;
        while (bs.pos < send) {
            const uint32_t token = bs_get(bs);
            uint32_t litlen = token >> 4;
            if (litlen == 15) {
                uint32_t b;
                do { b = bs_get(bs); litlen += b; } while (b == 255 && bs.pos < send);
            }
            if (bs.pos + litlen > send || op + litlen > dlen) { bad = true; break; }
            {   /* literal copy: 8 bytes per window peek, then byte tail */
                uint32_t left = litlen;
                while (left >= 8) {
                    uint64_t v = bs_peek64(bs);
                    bs.pos += 8;
                    sout[op + 0] = (uint8_t)v;       sout[op + 1] = (uint8_t)(v >> 8);
                    sout[op + 2] = (uint8_t)(v >> 16); sout[op + 3] = (uint8_t)(v >> 24);
                    sout[op + 4] = (uint8_t)(v >> 32); sout[op + 5] = (uint8_t)(v >> 40);
                    sout[op + 6] = (uint8_t)(v >> 48); sout[op + 7] = (uint8_t)(v >> 56);
                    op += 8;
                    left -= 8;
                }
                for (uint32_t j = 0; j < left; j++) sout[op + j] = bs_get(bs);
                op += left;
            }
            if (bs.pos >= send) break;       /* last sequence: literals only */

            if (bs.pos + 2 > send) { bad = true; break; }
            uint32_t offset = (uint32_t)bs_get(bs);
            offset |= (uint32_t)bs_get(bs) << 8;
            uint32_t mlen = token & 15;
            if (mlen == 15) {
                uint32_t b;
                do { b = bs_get(bs); mlen += b; } while (b == 255 && bs.pos < send);
            }
            mlen += 4;
            if (offset == 0 || offset > op || op + mlen > dlen) { bad = true; break; }
            /* uniform per-lane byte copy — control flow stays convergent
             * across the wave (a branchier word-copy variant measured 2x the
             * per-byte instructions from divergence; profiles/r01).
             * 4-wide manual unroll: the 4 loads are independent and issue
             * together instead of load-wait-store per byte. Overlap safety:
             * offset<4 would make loads of a quad depend on its own stores,
             * so quads only when offset>=4; sources are then complete. */
            const uint8_t *msrc = sout + op - offset;
            uint32_t j = 0;
            if (((offset | op) & 3) == 0) {
                /* word-granular stream (e.g. the 4-byte varlena slots of
                 * text columns): source, destination and offset are all
                 * 4-aligned, so copies run as aligned LDS words — the
                 * u32-splat case (offset 4: one repeated slot) is the
                 * dominant shape of low-cardinality flag columns */
                uint32_t *d32 = (uint32_t *)(sout + op);
                if (offset == 4) {
                    const uint32_t v = *(const uint32_t *)msrc;
                    for (; j + 4 <= mlen; j += 4) d32[j >> 2] = v;
                } else {
                    const uint32_t *s32 = (const uint32_t *)msrc;
                    const uint32_t woff = offset >> 2;
                    for (; j + 4 <= mlen; j += 4) {
                        /* overlap-safe: reads trail writes by woff words */
                        d32[j >> 2] = (j >> 2) < woff ? s32[j >> 2]
                                                      : d32[(j >> 2) - woff];
                    }
                }
            } else if (offset >= 8) {
                for (; j + 8 <= mlen; j += 8) {
                    uint8_t b0 = msrc[j], b1 = msrc[j + 1];
                    uint8_t b2 = msrc[j + 2], b3 = msrc[j + 3];
                    uint8_t b4 = msrc[j + 4], b5 = msrc[j + 5];
                    uint8_t b6 = msrc[j + 6], b7 = msrc[j + 7];
                    sout[op + j] = b0; sout[op + j + 1] = b1;
                    sout[op + j + 2] = b2; sout[op + j + 3] = b3;
                    sout[op + j + 4] = b4; sout[op + j + 5] = b5;
                    sout[op + j + 6] = b6; sout[op + j + 7] = b7;
                }
            } else if (offset >= 4) {
                for (; j + 4 <= mlen; j += 4) {
                    uint8_t b0 = msrc[j], b1 = msrc[j + 1];
                    uint8_t b2 = msrc[j + 2], b3 = msrc[j + 3];
                    sout[op + j] = b0; sout[op + j + 1] = b1;
                    sout[op + j + 2] = b2; sout[op + j + 3] = b3;
                }
            }
            for (; j < mlen; j++) sout[op + j] = msrc[j];
            op += mlen;
        }
        if (bad || op != dlen) atomicOr(err, 4);
    }
}

__global__ void lz4_decode_lane_kernel(
    const uint8_t *__restrict__ data, uint8_t *__restrict__ scratch,
    const SegDesc *__restrict__ segs, uint32_t n_segs, uint32_t stride,
    int *__restrict__ err)
{
    extern __shared__ uint8_t sout_all[];
    const uint32_t tid = threadIdx.x;
    const uint32_t first = blockIdx.x * blockDim.x;
    const uint32_t gseg = first + tid;
    uint8_t *sout = sout_all + (size_t)tid * stride;
    /* per-segment (dst_off, decomp_len) cached in LDS for the flush loop */
    uint64_t *sdst = (uint64_t *)(sout_all + (size_t)blockDim.x * stride);
    uint32_t *sdlen = (uint32_t *)(sdst + blockDim.x);

    if (gseg < n_segs) {
        const SegDesc s = segs[gseg];
        sdst[tid] = s.dst_off;
        sdlen[tid] = s.decomp_len;
        lz4_lane_decode(data, s, sout, err);
    }
    __syncthreads();

    /* cooperative coalesced flush: 16 B global stores (full coalescing)
     * assembled from four 4 B LDS reads — the region stride is a 4-mod-8
     * multiple so the decode loop's byte ops are bank-conflict-free, which
     * makes LDS-side 16 B alignment unavailable */
    const uint32_t eps = (stride + 15) >> 4;          /* 16B slots per segment */
    const uint32_t in_block = min(blockDim.x, n_segs - first);
    const uint32_t total = in_block * eps;
    for (uint32_t f = tid; f < total; f += blockDim.x) {
        const uint32_t sidx = f / eps;
        const uint32_t boff = (f % eps) << 4;
        const uint32_t dl = sdlen[sidx];
        if (boff >= dl) continue;
        uint8_t *dst = scratch + sdst[sidx] + boff;
        const uint32_t *ls = (const uint32_t *)(sout_all + (size_t)sidx * stride + boff);
        if (boff + 16 <= dl) {
            uint4 v;
            v.x = ls[0]; v.y = ls[1]; v.z = ls[2]; v.w = ls[3];
            *(uint4 *)dst = v;                        /* dst 16B-aligned (writer) */
        } else {
            const uint8_t *lb = (const uint8_t *)ls;
            for (uint32_t j = 0; j < dl - boff; j++) dst[j] = lb[j];
        }
    }
}

__global__ __launch_bounds__(WAVE) void lz4_decode_lds_kernel(
    const uint8_t *__restrict__ data, uint8_t *__restrict__ scratch,
    const SegDesc *__restrict__ segs, uint32_t in_cap, int *__restrict__ err)
{
    extern __shared__ uint8_t sbuf[];      /* [in_cap compressed][output] */
    const SegDesc s = segs[blockIdx.x];
    const uint32_t lane = threadIdx.x;

    /* stage compressed bytes: aligned u32 loads from (src_off & ~3) */
    const uint64_t base = s.src_off & ~3ull;
    const uint32_t shift = (uint32_t)(s.src_off - base);
    const uint32_t words = (s.comp_len + shift + 3) >> 2;
    const uint32_t *gsrc = (const uint32_t *)(data + base);
    uint32_t *lin = (uint32_t *)sbuf;
    for (uint32_t j = lane; j < words; j += WAVE) lin[j] = gsrc[j];
    __syncthreads();

    uint8_t *out = sbuf + in_cap;
    lz4_body<true>(sbuf + shift, s.comp_len, out, s.decomp_len, lane, err);
    __syncthreads();

    /* coalesced flush LDS -> global */
    uint8_t *dst = scratch + s.dst_off;
    const uint32_t dlen = s.decomp_len;
    if (((uintptr_t)dst & 15) == 0) {
        uint32_t vec = dlen >> 4;
        uint4 *d4 = (uint4 *)dst;
        const uint4 *o4 = (const uint4 *)out;
        for (uint32_t j = lane; j < vec; j += WAVE) d4[j] = o4[j];
        for (uint32_t j = (vec << 4) + lane; j < dlen; j += WAVE) dst[j] = out[j];
    } else {
        for (uint32_t j = lane; j < dlen; j += WAVE) dst[j] = out[j];
    }
}

/* =====================================================================
 * Fused filter + partial aggregate (generic, ungrouped)
 * grid: n_groups * tiles_per_group blocks of AGG_BLOCK threads.
 * Each thread strides rows of its tile; per-thread register accumulators;
 * wave __shfl_down reduce; LDS cross-wave reduce; one AccCell per block.
 * ===================================================================== */

/* canonical-stream byte position of value j's low-L bytes, by mode family:
 * lz4 P(L) interleaves them with match tokens (csf_canon_p_pos); the
 * restricted-zstd P(L) frame holds them densely in its literal section
 * (csf_canon_zrp_pos) */
__device__ inline uint32_t canon_pos(uint8_t mode, uint32_t L, uint32_t j)
{
    if ((mode & 0xF0u) == CSF_SEGMODE_ZRP_BASE)
        return j == 0 ? 15u : 23u + (j - 1u) * L;
    const uint32_t pos = j * (L + 3u) + (6u - L);
    return j == 0 ? 1u : (j == 1 ? 9u : pos);
}

__device__ inline bool col_value(const uint8_t *__restrict__ data,
                                 const uint8_t *__restrict__ scratch,
                                 const uint32_t *__restrict__ rank,
                                 const ColLoc &cl, uint32_t row,
                                 int64_t &iv, double &fv)
{
    uint32_t idx = row;
    if (!(cl.flags & 2)) {   /* sparse: null check + rank indirection */
        const uint64_t *bm = (const uint64_t *)(data + cl.exists_off);
        uint64_t w = bm[row >> 6];
        uint64_t bit = 1ull << (row & 63);
        if (!(w & bit)) return false;
        idx = rank[cl.rank_off + (row >> 6)] + (uint32_t)__popcll(w & (bit - 1));
    }
    const uint8_t *base = (cl.flags & 1) ? scratch + cl.val_off : data + cl.val_off;
    if (cl.flags & 4) {
        /* canonical stream: value bytes at a CLOSED-FORM position in the
         * compressed stream — no LZ4 sequence parse, no scratch (format.h).
         * One unaligned dwordx2 load per value; consecutive rows sit
         * (L+3) bytes apart, so a wave's loads coalesce into a few lines. */
        if (cl.mode == CSF_SEGMODE_LIT) {
            base += (uint64_t)cl.hval;           /* skip literal-run header */
        } else if ((cl.mode & 0xF8u) == CSF_SEGMODE_ZR4B_BASE) {
            /* width-4 single-varying-byte / const slots (char(1) flags) */
            uint32_t slot = (uint32_t)cl.hval;
            if (cl.mode != CSF_SEGMODE_ZR4_CONST) {
                const uint32_t kk = cl.L;
                const uint32_t pos = 15u + (idx == 0 ? kk : idx + 3u + kk);
                slot |= (uint32_t)base[pos] << (8u * kk);
            }
            if (cl.type == CSTRIPE_F32) {
                float ff;
                __builtin_memcpy(&ff, &slot, 4);
                fv = ff;
                iv = 0;
            } else {
                iv = cl.type == CSTRIPE_TEXT ? (int64_t)slot : (int64_t)(int32_t)slot;
                fv = cl.type == CSTRIPE_TEXT ? 0.0 : (double)iv;
            }
            return true;
        } else {
            uint64_t raw;
            if (cl.mode == CSF_SEGMODE_CONST || cl.mode == CSF_SEGMODE_ZR_CONST) {
                raw = (uint64_t)cl.hval;
            } else {                             /* P(L), lz4 or zstd layout */
                const uint32_t Lx = cl.L;
                const uint32_t pos = canon_pos(cl.mode, Lx, idx);
                uint64_t lo;
                __builtin_memcpy(&lo, base + pos, 8);
                const uint64_t m = (~0ull) >> ((8u - Lx) * 8u);
                raw = (lo & m) | (uint64_t)cl.hval;
            }
            if (cl.type == CSTRIPE_F64) {
                __builtin_memcpy(&fv, &raw, 8);
                iv = 0;
            } else {
                iv = (int64_t)raw;
                fv = (double)iv;
            }
            return true;
        }
    }
    switch (cl.type) {
        case CSTRIPE_I8:  iv = ((const int8_t  *)base)[idx]; fv = (double)iv; break;
        case CSTRIPE_I16: iv = ((const int16_t *)base)[idx]; fv = (double)iv; break;
        case CSTRIPE_I32: iv = ((const int32_t *)base)[idx]; fv = (double)iv; break;
        case CSTRIPE_I64: iv = ((const int64_t *)base)[idx]; fv = (double)iv; break;
        case CSTRIPE_F32: fv = ((const float  *)base)[idx]; iv = 0; break;
        case CSTRIPE_TEXT: iv = (int64_t)((const uint32_t *)base)[idx]; fv = 0; break;
        default:          fv = ((const double *)base)[idx]; iv = 0; break;
    }
    return true;
}

/* PG float ordering (float.c float8_cmp_internal): NaN sorts greater than
 * every non-NaN and equal to itself — predicate + MIN/MAX semantics must
 * match the reference's float8 operators, not IEEE compares */
__device__ inline int f64cmp_pg(double a, double b)
{
    if (a > b) return 1;
    if (a < b) return -1;
    if (a == b) return 0;
    const bool na = (a != a), nb = (b != b);
    if (na && nb) return 0;
    return na ? 1 : -1;
}

__device__ inline bool pred_eval(const PredD &p, int64_t iv, double fv)
{
    if (p.is_float) {
        const int c = f64cmp_pg(fv, p.fval);
        switch (p.op) {
            case CSTRIPE_PRED_LT: return c <  0;
            case CSTRIPE_PRED_LE: return c <= 0;
            case CSTRIPE_PRED_GT: return c >  0;
            case CSTRIPE_PRED_GE: return c >= 0;
            case CSTRIPE_PRED_EQ: return c == 0;
            default:              return c != 0;
        }
    }
    switch (p.op) {
        case CSTRIPE_PRED_LT: return iv <  p.ival;
        case CSTRIPE_PRED_LE: return iv <= p.ival;
        case CSTRIPE_PRED_GT: return iv >  p.ival;
        case CSTRIPE_PRED_GE: return iv >= p.ival;
        case CSTRIPE_PRED_EQ: return iv == p.ival;
        default:              return iv != p.ival;
    }
}

struct ThreadAcc {
    int64_t lo, hi;   /* i128 sum (lo unsigned semantics) / i64 minmax in lo */
    double  f;
    int64_t cnt;
};

__device__ inline void acc_init(ThreadAcc &a, uint8_t kind)
{
    a.lo = 0; a.hi = 0; a.f = 0.0; a.cnt = 0;
    switch (kind) {
        case CSTRIPE_AGG_MIN_I64: a.lo = INT64_MAX; break;
        case CSTRIPE_AGG_MAX_I64: a.lo = INT64_MIN; break;
        /* PG order: NaN is the greatest float, so NaN is MIN's identity
         * (min(NaN, x) = x; min over only-NaN rows correctly stays NaN) */
        case CSTRIPE_AGG_MIN_F64: a.f = NAN; break;
        case CSTRIPE_AGG_MAX_F64: a.f = -INFINITY; break;
        default: break;
    }
}

__device__ inline void acc_add_i128(ThreadAcc &a, __int128 v)
{
    __int128 x = ((__int128)a.hi << 64) | (unsigned long long)a.lo;
    x += v;
    a.lo = (int64_t)(uint64_t)x;
    a.hi = (int64_t)(x >> 64);
}

__device__ inline void acc_merge(ThreadAcc &a, const ThreadAcc &b, uint8_t kind)
{
    switch (kind) {
        case CSTRIPE_AGG_MIN_I64: a.lo = min(a.lo, b.lo); break;
        case CSTRIPE_AGG_MAX_I64: a.lo = max(a.lo, b.lo); break;
        case CSTRIPE_AGG_MIN_F64: if (f64cmp_pg(b.f, a.f) < 0) a.f = b.f; break;
        case CSTRIPE_AGG_MAX_F64: if (f64cmp_pg(b.f, a.f) > 0) a.f = b.f; break;
        case CSTRIPE_AGG_SUM_F64: a.f += b.f; break;
        default: {  /* counts + i128 sums */
            unsigned long long lo = (unsigned long long)a.lo + (unsigned long long)b.lo;
            int64_t carry = lo < (unsigned long long)a.lo ? 1 : 0;
            a.lo = (int64_t)lo;
            a.hi = a.hi + b.hi + carry;
            break;
        }
    }
    a.cnt += b.cnt;
}

__device__ inline void acc_row(ThreadAcc &a, const AggD &g,
                               const uint8_t *__restrict__ data,
                               const uint8_t *__restrict__ scratch,
                               const uint32_t *__restrict__ rank,
                               const ColLoc *__restrict__ cols, uint32_t row,
                               bool pass)
{
    /* predicated accumulate: operand loads issue unconditionally (so row
     * iterations pipeline with no divergent skip), the fold is masked by
     * `pass` — the row loop in filter_agg_kernel has no continue */
    int64_t iv; double fv;
    switch (g.kind) {
        case CSTRIPE_AGG_COUNT_STAR:
            if (pass) { a.cnt++; a.lo++; }
            break;
        case CSTRIPE_AGG_COUNT_COL: {
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            if (ok & pass) { a.cnt++; a.lo++; }
            break;
        }
        case CSTRIPE_AGG_SUM_I64: {
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            if (ok & pass) {
                acc_add_i128(a, (__int128)iv);
                a.cnt++;
            }
            break;
        }
        case CSTRIPE_AGG_SUM_F64: {
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            if (ok & pass) { a.f += fv; a.cnt++; }
            break;
        }
        case CSTRIPE_AGG_MIN_I64: {
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            if (ok & pass) { a.lo = min(a.lo, iv); a.cnt++; }
            break;
        }
        case CSTRIPE_AGG_MAX_I64: {
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            if (ok & pass) { a.lo = max(a.lo, iv); a.cnt++; }
            break;
        }
        case CSTRIPE_AGG_MIN_F64: {
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            if (ok & pass) {
                if (f64cmp_pg(fv, a.f) < 0) a.f = fv;
                a.cnt++;
            }
            break;
        }
        case CSTRIPE_AGG_MAX_F64: {
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            if (ok & pass) {
                if (f64cmp_pg(fv, a.f) > 0) a.f = fv;
                a.cnt++;
            }
            break;
        }
        case CSTRIPE_AGG_SUM_PROD_I64: {
            int64_t ib; double fb;
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            ok = ok & col_value(data, scratch, rank, cols[g.proj_b], row, ib, fb);
            if (ok & pass) {
                acc_add_i128(a, (__int128)iv * ib);
                a.cnt++;
            }
            break;
        }
        case CSTRIPE_AGG_SUM_DISC_I64: {
            int64_t ib; double fb;
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            ok = ok & col_value(data, scratch, rank, cols[g.proj_b], row, ib, fb);
            if (ok & pass) {
                acc_add_i128(a, (__int128)iv * (g.one - ib));
                a.cnt++;
            }
            break;
        }
        case CSTRIPE_AGG_SUM_DISC_TAX_I64: {
            int64_t ib, ic; double fb, fc;
            bool ok = col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv);
            ok = ok & col_value(data, scratch, rank, cols[g.proj_b], row, ib, fb);
            ok = ok & col_value(data, scratch, rank, cols[g.proj_c], row, ic, fc);
            if (ok & pass) {
                acc_add_i128(a, (__int128)iv * (g.one - ib) * (g.one + ic));
                a.cnt++;
            }
            break;
        }
    }
}

__device__ inline void wave_reduce(ThreadAcc &a, uint8_t kind)
{
    for (int d = WAVE / 2; d > 0; d >>= 1) {
        ThreadAcc b;
        b.lo  = __shfl_down((long long)a.lo, d, WAVE);
        b.hi  = __shfl_down((long long)a.hi, d, WAVE);
        b.f   = __shfl_down(a.f, d, WAVE);
        b.cnt = __shfl_down((long long)a.cnt, d, WAVE);
        acc_merge(a, b, kind);
    }
}

template <int NPREDS, int NAGGS>
__global__ __launch_bounds__(AGG_BLOCK) void filter_agg_kernel(
    const uint8_t *__restrict__ data, const uint8_t *__restrict__ scratch,
    const uint32_t *__restrict__ rank, const GroupDesc *__restrict__ groups,
    const ColLoc *__restrict__ colloc, AccCell *__restrict__ block_out,
    const AggParams params)
{
    /* compile-time trip counts (<0 = runtime) unroll the pred/agg loops so
     * every load is straight-line and pipelines (cdna guide §5.4 trap c) */
    const uint32_t n_preds = NPREDS >= 0 ? (uint32_t)NPREDS : params.n_preds;
    const uint32_t n_aggs = NAGGS >= 0 ? (uint32_t)NAGGS : params.n_aggs;
    const uint32_t gid = blockIdx.x / params.tiles_per_group;
    const uint32_t tile = blockIdx.x % params.tiles_per_group;
    const GroupDesc g = groups[gid];
    const ColLoc *cols = colloc + g.colbase;

    uint32_t row_start = tile * TILE_ROWS;
    uint32_t row_end = min(row_start + TILE_ROWS, g.row_count);

    ThreadAcc acc[NAGGS >= 0 ? NAGGS : MAX_AGGS];
    #pragma unroll
    for (uint32_t a = 0; a < n_aggs; a++) acc_init(acc[a], params.aggs[a].kind);

    for (uint32_t row = row_start + threadIdx.x; row < row_end; row += AGG_BLOCK) {
        bool pass = true, gv = false;
        int last_proj = -1;
        int64_t liv = 0; double lfv = 0; bool lok = false;
        /* no short-circuit: loads stay control-independent so they issue
         * back-to-back and pipeline instead of chaining load->wait->branch.
         * contiguous OR groups accumulate into gv, folded at group end */
        #pragma unroll
        for (uint32_t p = 0; p < n_preds; p++) {
            const PredD &pr = params.preds[p];
            if ((int)pr.proj != last_proj) {   /* BETWEEN reuses the load */
                lok = col_value(data, scratch, rank, cols[pr.proj], row, liv, lfv);
                last_proj = (int)pr.proj;
            }
            gv = gv | (lok && pred_eval(pr, liv, lfv));
            if (pr.gend) { pass = pass & gv; gv = false; }
        }
        if (!pass) continue;
        #pragma unroll
        for (uint32_t a = 0; a < n_aggs; a++)
            acc_row(acc[a], params.aggs[a], data, scratch, rank, cols, row, true);
    }

    /* wave reduce then cross-wave via LDS */
    __shared__ ThreadAcc lds[AGG_BLOCK / WAVE][MAX_AGGS];
    const uint32_t wid = threadIdx.x / WAVE;
    const uint32_t lane = threadIdx.x % WAVE;
    #pragma unroll
    for (uint32_t a = 0; a < n_aggs; a++) {
        wave_reduce(acc[a], params.aggs[a].kind);
        if (lane == 0) lds[wid][a] = acc[a];
    }
    __syncthreads();
    if (wid == 0) {
        for (uint32_t a = lane; a < n_aggs; a += WAVE) {
            ThreadAcc r = lds[0][a];
            for (uint32_t w = 1; w < AGG_BLOCK / WAVE; w++)
                acc_merge(r, lds[w][a], params.aggs[a].kind);
            AccCell c;
            c.lo = r.lo; c.hi = r.hi; c.f = r.f; c.cnt = r.cnt;
            block_out[(uint64_t)blockIdx.x * n_aggs + a] = c;
        }
    }
}


/* =====================================================================
 * FUSED decode -> filter -> partial aggregate: one block
 * decodes a 2048-row tile of ALL projected columns into LDS (one lane per
 * 256 B segment — 64 segments/column/tile, up to 4 i64 columns = 256 lanes)
 * and aggregates straight from LDS. The decompressed stream never touches
 * HBM: eliminates the scratch write + re-read of the two-kernel path.
 * Eligible when every projected column is dense int64 in uniform 256 B LZ4
 * segments (the writer default); otherwise the two-kernel path runs.
 * ===================================================================== */

#define FUSE_TILE_ROWS 2048
#define FUSE_STRIDE 280          /* 8B-aligned per-lane region (val() u64 reads) */

struct FusedTile {
    uint32_t row_count;          /* rows in this tile (<= FUSE_TILE_ROWS) */
    uint32_t seg_base[4];        /* d_segs index of tile's first segment per proj col */
};

template <int NPREDS, int NAGGS>
__global__ __launch_bounds__(AGG_BLOCK) void fused_agg_kernel(
    const uint8_t *__restrict__ data, const SegDesc *__restrict__ segs,
    const FusedTile *__restrict__ tiles, AccCell *__restrict__ block_out,
    int *__restrict__ err, const AggParams params)
{
    const uint32_t n_preds = NPREDS >= 0 ? (uint32_t)NPREDS : params.n_preds;
    const uint32_t n_aggs = NAGGS >= 0 ? (uint32_t)NAGGS : params.n_aggs;
    extern __shared__ uint8_t lds[];                 /* 256 * FUSE_STRIDE */
    const uint32_t tid = threadIdx.x;
    const FusedTile t = tiles[blockIdx.x];

    /* phase 1: decode my segment (lane = col*64 + seg_in_tile) */
    const uint32_t c = tid >> 6;
    const uint32_t sseg = tid & 63;
    const uint32_t nsegs = (t.row_count * 8 + 255) >> 8;
    if (c < params.n_proj && sseg < nsegs) {
        const SegDesc sd = segs[t.seg_base[c] + sseg];
        lz4_lane_decode(data, sd, lds + (size_t)tid * FUSE_STRIDE, err);
    }
    __syncthreads();

    /* phase 2: filter + aggregate from LDS */
    ThreadAcc acc[NAGGS >= 0 ? NAGGS : MAX_AGGS];
    #pragma unroll
    for (uint32_t a = 0; a < n_aggs; a++) acc_init(acc[a], params.aggs[a].kind);

    auto val = [&](uint32_t proj, uint32_t row) -> int64_t {
        const uint32_t region = proj * 64 + (row >> 5);
        return *(const int64_t *)(lds + (size_t)region * FUSE_STRIDE + ((row & 31) << 3));
    };

    for (uint32_t row = tid; row < t.row_count; row += AGG_BLOCK) {
        bool pass = true, gv = false;
        int last_proj = -1;
        int64_t liv = 0;
        for (uint32_t p = 0; p < n_preds; p++) {
            const PredD &pr = params.preds[p];
            if ((int)pr.proj != last_proj) { liv = val(pr.proj, row); last_proj = (int)pr.proj; }
            gv = gv | pred_eval(pr, liv, 0.0);       /* cols are i64: is_float==0 */
            if (pr.gend) { pass = pass & gv; gv = false; }
        }
        if (!pass) continue;
        #pragma unroll
        for (uint32_t a = 0; a < n_aggs; a++) {
            const AggD &g = params.aggs[a];
            ThreadAcc &A = acc[a];
            switch (g.kind) {
                case CSTRIPE_AGG_COUNT_STAR:
                case CSTRIPE_AGG_COUNT_COL:          /* dense: col always present */
                    A.cnt++; A.lo++; break;
                case CSTRIPE_AGG_SUM_I64:
                    acc_add_i128(A, (__int128)val(g.proj_a, row)); A.cnt++; break;
                case CSTRIPE_AGG_MIN_I64:
                    A.lo = min(A.lo, val(g.proj_a, row)); A.cnt++; break;
                case CSTRIPE_AGG_MAX_I64:
                    A.lo = max(A.lo, val(g.proj_a, row)); A.cnt++; break;
                case CSTRIPE_AGG_SUM_PROD_I64:
                    acc_add_i128(A, (__int128)val(g.proj_a, row) * val(g.proj_b, row));
                    A.cnt++; break;
                case CSTRIPE_AGG_SUM_DISC_I64:
                    acc_add_i128(A, (__int128)val(g.proj_a, row) * (g.one - val(g.proj_b, row)));
                    A.cnt++; break;
                case CSTRIPE_AGG_SUM_DISC_TAX_I64:
                    acc_add_i128(A, (__int128)val(g.proj_a, row) * (g.one - val(g.proj_b, row))
                                     * (g.one + val(g.proj_c, row)));
                    A.cnt++; break;
                default: break;                       /* host guards float kinds */
            }
        }
    }

    /* reduce: wave shuffle -> cross-wave LDS -> one AccCell per block */
    __shared__ ThreadAcc lred[AGG_BLOCK / WAVE][MAX_AGGS];
    const uint32_t wid = tid / WAVE;
    const uint32_t lane = tid % WAVE;
    #pragma unroll
    for (uint32_t a = 0; a < n_aggs; a++) {
        wave_reduce(acc[a], params.aggs[a].kind);
        if (lane == 0) lred[wid][a] = acc[a];
    }
    __syncthreads();
    if (wid == 0) {
        for (uint32_t a = lane; a < n_aggs; a += WAVE) {
            ThreadAcc r = lred[0][a];
            for (uint32_t w = 1; w < AGG_BLOCK / WAVE; w++)
                acc_merge(r, lred[w][a], params.aggs[a].kind);
            AccCell cell;
            cell.lo = r.lo; cell.hi = r.hi; cell.f = r.f; cell.cnt = r.cnt;
            block_out[(uint64_t)blockIdx.x * n_aggs + a] = cell;
        }
    }
}

/* ---- R-row variant (R=8): one (R-1)*step+8 <= 57 B window per column.
 * step in {4..7} is wave-uniform per column, so a 4-way uniform branch gives
 * each extraction COMPILE-TIME shift amounts (no dynamic register indexing,
 * no scratch). Doubles in-flight loads again over the quad kernel. ---- */

/* raw per-row values: int columns store the integer, float columns store
 * the double's BIT PATTERN (converted only at use) — halves the register
 * footprint of a staged row group, which is what caps occupancy here */
template <int R>
struct ValsR {
    int64_t v[R];
    __device__ inline double fget(int k) const {
        double d;
        __builtin_memcpy(&d, &v[k], 8);
        return d;
    }
};

template <int R, uint32_t STEP>
__device__ inline void canon_extract(const uint8_t *__restrict__ sp,
                                     uint32_t pos, uint64_t m, uint64_t hval,
                                     uint64_t (&r)[R])
{
    /* per-u64 window loads (measured faster than 16 B dwordx4 quads here:
     * the unaligned quads span more lines per instruction and raised the
     * kernel's spill pressure for no latency win) */
    constexpr uint32_t NW = ((R - 1) * STEP + 8 + 7) / 8;
    uint64_t w[NW];
    #pragma unroll
    for (uint32_t i = 0; i < NW; i++)
        __builtin_memcpy(&w[i], sp + pos + 8 * i, 8);
    #pragma unroll
    for (uint32_t k = 0; k < R; k++) {
        const uint32_t d = k * STEP;
        const uint32_t wi = d >> 3, sh = (d & 7u) * 8u;
        uint64_t raw = sh ? ((w[wi] >> sh) | (w[wi + 1 < NW ? wi + 1 : wi] << (64u - sh)))
                          : w[wi];
        r[k] = (raw & m) | hval;
    }
}

template <int R>
__device__ inline void col_multi(const uint8_t *__restrict__ data,
                                 const uint8_t *__restrict__ scratch,
                                 const ColLoc &cl, uint32_t row, ValsR<R> &q)
{
    const uint8_t *base = (cl.flags & 1) ? scratch + cl.val_off : data + cl.val_off;
    if ((cl.flags & 4) && cl.mode != CSF_SEGMODE_LIT) {
        if ((cl.mode & 0xF8u) == CSF_SEGMODE_ZR4B_BASE) {
            /* width-4 single-varying-byte / const slots (char(1) flags):
             * ONE u64 window load covers 8 rows' varying bytes (stride 1) */
            const uint32_t kk = cl.L;
            uint32_t slots[R];
            if (cl.mode == CSF_SEGMODE_ZR4_CONST) {
                #pragma unroll
                for (int k = 0; k < R; k++) slots[k] = (uint32_t)cl.hval;
            } else if (row == 0) {
                #pragma unroll
                for (int k = 0; k < R; k++) {
                    const uint32_t j = row + (uint32_t)k;
                    const uint32_t pos = 15u + (j == 0 ? kk : j + 3u + kk);
                    slots[k] = (uint32_t)cl.hval |
                               ((uint32_t)base[pos] << (8u * kk));
                }
            } else {
                uint64_t w[(R + 7) / 8];
                #pragma unroll
                for (int i = 0; i < (R + 7) / 8; i++)
                    __builtin_memcpy(&w[i], base + 15u + row + 3u + kk + 8u * i, 8);
                #pragma unroll
                for (int k = 0; k < R; k++)
                    slots[k] = (uint32_t)cl.hval |
                               ((uint32_t)((w[k >> 3] >> (8 * (k & 7))) & 0xFFu)
                                << (8u * kk));
            }
            #pragma unroll
            for (int k = 0; k < R; k++) {
                if (cl.type == CSTRIPE_F32) {
                    float ff;
                    __builtin_memcpy(&ff, &slots[k], 4);
                    const double d = ff;
                    __builtin_memcpy(&q.v[k], &d, 8);
                } else if (cl.type == CSTRIPE_TEXT) {
                    q.v[k] = (int64_t)slots[k];
                } else {
                    q.v[k] = (int64_t)(int32_t)slots[k];
                }
            }
            return;
        }
        uint64_t r[R];
        if (cl.mode == CSF_SEGMODE_CONST || cl.mode == CSF_SEGMODE_ZR_CONST) {
            #pragma unroll
            for (int k = 0; k < R; k++) r[k] = (uint64_t)cl.hval;
        } else if ((cl.mode & 0xF0u) == CSF_SEGMODE_ZRP_BASE) {
            /* zstd P(L): dense low-L bytes in the literal section, byte
             * stride L (canon_pos); only j=0 is special */
            const uint32_t Lx = cl.L;
            const uint64_t m = (~0ull) >> ((8u - Lx) * 8u);
            if (row == 0) {
                #pragma unroll
                for (int k = 0; k < R; k++) {
                    uint64_t a;
                    __builtin_memcpy(&a, base + canon_pos(cl.mode, Lx, row + k), 8);
                    r[k] = (a & m) | (uint64_t)cl.hval;
                }
            } else {
                const uint32_t pos = 23u + (row - 1u) * Lx;
                switch (Lx) {                 /* wave-uniform */
                    case 1: canon_extract<R, 1>(base, pos, m, (uint64_t)cl.hval, r); break;
                    case 2: canon_extract<R, 2>(base, pos, m, (uint64_t)cl.hval, r); break;
                    case 3: canon_extract<R, 3>(base, pos, m, (uint64_t)cl.hval, r); break;
                    default: canon_extract<R, 4>(base, pos, m, (uint64_t)cl.hval, r); break;
                }
            }
        } else {                              /* lz4 P(L) */
            const uint32_t Lx = cl.L, step = Lx + 3u;
            const uint64_t m = (~0ull) >> ((8u - Lx) * 8u);
            if (row < 2) {                    /* j=0 -> 1, j=1 -> 9 specials */
                #pragma unroll
                for (int k = 0; k < R; k++) {
                    const uint32_t j = row + k;
                    uint32_t pos = j * step + (6u - Lx);
                    pos = j == 0 ? 1u : (j == 1 ? 9u : pos);
                    uint64_t a;
                    __builtin_memcpy(&a, base + pos, 8);
                    r[k] = (a & m) | (uint64_t)cl.hval;
                }
            } else {
                const uint32_t pos = row * step + (6u - Lx);
                /* wave-uniform branch on step -> compile-time shifts */
                switch (step) {
                    case 4: canon_extract<R, 4>(base, pos, m, (uint64_t)cl.hval, r); break;
                    case 5: canon_extract<R, 5>(base, pos, m, (uint64_t)cl.hval, r); break;
                    case 6: canon_extract<R, 6>(base, pos, m, (uint64_t)cl.hval, r); break;
                    default: canon_extract<R, 7>(base, pos, m, (uint64_t)cl.hval, r); break;
                }
            }
        }
        #pragma unroll
        for (int k = 0; k < R; k++) q.v[k] = (int64_t)r[k];   /* F64: bits */
        return;
    }
    if (cl.flags & 4) base += (uint64_t)cl.hval;     /* LIT header */
    switch (cl.type) {                                /* row % R == 0 */
        case CSTRIPE_I8: {
            const int8_t *p = (const int8_t *)base + row;
            #pragma unroll
            for (int k = 0; k < R; k++) q.v[k] = p[k];
            break;
        }
        case CSTRIPE_I16: {
            const int16_t *p = (const int16_t *)base + row;
            #pragma unroll
            for (int k = 0; k < R; k++) q.v[k] = p[k];
            break;
        }
        case CSTRIPE_I32: {
            const int32_t *p = (const int32_t *)base + row;
            #pragma unroll
            for (int k = 0; k < R; k++) q.v[k] = p[k];
            break;
        }
        case CSTRIPE_I64: {
            const uint8_t *p = base + (size_t)row * 8;
            #pragma unroll
            for (int k = 0; k < R; k += 2) {
                struct { int64_t a, b; } t;
                __builtin_memcpy(&t, p + 16 * (k / 2), 16);
                q.v[k] = t.a;
                if (k + 1 < R) q.v[k + 1] = t.b;
            }
            break;
        }
        case CSTRIPE_F32: {
            const float *p = (const float *)base + row;
            #pragma unroll
            for (int k = 0; k < R; k++) {
                const double d = p[k];
                __builtin_memcpy(&q.v[k], &d, 8);     /* widen, store bits */
            }
            break;
        }
        case CSTRIPE_TEXT: {
            const uint32_t *p = (const uint32_t *)base + row;
            #pragma unroll
            for (int k = 0; k < R; k++) q.v[k] = (int64_t)p[k];
            break;
        }
        default: {                                    /* F64 bits */
            const uint8_t *p = base + (size_t)row * 8;
            #pragma unroll
            for (int k = 0; k < R; k += 2) {
                struct { int64_t a, b; } t;
                __builtin_memcpy(&t, p + 16 * (k / 2), 16);
                q.v[k] = t.a;
                if (k + 1 < R) q.v[k + 1] = t.b;
            }
            break;
        }
    }
}

template <int R>
__device__ inline void acc_multi(ThreadAcc &a, const AggD &g,
                                 const uint8_t *__restrict__ data,
                                 const uint8_t *__restrict__ scratch,
                                 const ColLoc *__restrict__ cols, uint32_t row,
                                 const bool (&pv)[R])
{
    ValsR<R> qa, qb, qc;
    int64_t c = 0;
    #pragma unroll
    for (int k = 0; k < R; k++) c += (int)pv[k];
    switch (g.kind) {
        case CSTRIPE_AGG_COUNT_STAR:
        case CSTRIPE_AGG_COUNT_COL:
            a.cnt += c;
            a.lo += c;
            return;
        case CSTRIPE_AGG_SUM_I64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k]) acc_add_i128(a, (__int128)qa.v[k]);
            break;
        case CSTRIPE_AGG_SUM_F64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k]) a.f += qa.fget(k);
            break;
        case CSTRIPE_AGG_MIN_I64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k]) a.lo = min(a.lo, qa.v[k]);
            break;
        case CSTRIPE_AGG_MAX_I64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k]) a.lo = max(a.lo, qa.v[k]);
            break;
        case CSTRIPE_AGG_MIN_F64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k] && f64cmp_pg(qa.fget(k), a.f) < 0) a.f = qa.fget(k);
            break;
        case CSTRIPE_AGG_MAX_F64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k] && f64cmp_pg(qa.fget(k), a.f) > 0) a.f = qa.fget(k);
            break;
        case CSTRIPE_AGG_SUM_PROD_I64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            col_multi<R>(data, scratch, cols[g.proj_b], row, qb);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k]) acc_add_i128(a, (__int128)qa.v[k] * qb.v[k]);
            break;
        case CSTRIPE_AGG_SUM_DISC_I64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            col_multi<R>(data, scratch, cols[g.proj_b], row, qb);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k]) acc_add_i128(a, (__int128)qa.v[k] * (g.one - qb.v[k]));
            break;
        case CSTRIPE_AGG_SUM_DISC_TAX_I64:
            col_multi<R>(data, scratch, cols[g.proj_a], row, qa);
            col_multi<R>(data, scratch, cols[g.proj_b], row, qb);
            col_multi<R>(data, scratch, cols[g.proj_c], row, qc);
            #pragma unroll
            for (int k = 0; k < R; k++)
                if (pv[k]) acc_add_i128(a, (__int128)qa.v[k] * (g.one - qb.v[k])
                                           * (g.one + qc.v[k]));
            break;
        default:
            break;
    }
    a.cnt += c;
}

template <int NPREDS, int NAGGS, int R, int MINW = 1>
__global__ __launch_bounds__(AGG_BLOCK, MINW) void multi_agg_kernel(
    const uint8_t *__restrict__ data, const uint8_t *__restrict__ scratch,
    const GroupDesc *__restrict__ groups, const ColLoc *__restrict__ colloc,
    AccCell *__restrict__ block_out, const AggParams params)
{
    const uint32_t n_preds = NPREDS >= 0 ? (uint32_t)NPREDS : params.n_preds;
    const uint32_t n_aggs = NAGGS >= 0 ? (uint32_t)NAGGS : params.n_aggs;
    const uint32_t gid = blockIdx.x / params.tiles_per_group;
    const uint32_t tile = blockIdx.x % params.tiles_per_group;
    const GroupDesc g = groups[gid];
    const ColLoc *cols = colloc + g.colbase;

    const uint32_t row_start = tile * TILE_ROWS;
    const uint32_t row_end = min(row_start + TILE_ROWS, g.row_count);

    ThreadAcc acc[NAGGS >= 0 ? NAGGS : MAX_AGGS];
    #pragma unroll
    for (uint32_t a = 0; a < n_aggs; a++) acc_init(acc[a], params.aggs[a].kind);

    for (uint32_t row = row_start + R * threadIdx.x; row < row_end;
         row += R * AGG_BLOCK) {
        bool pv[R], gv[R];
        #pragma unroll
        for (int k = 0; k < R; k++) { pv[k] = row + k < row_end; gv[k] = false; }
        int last_proj = -1;
        ValsR<R> q{};
        #pragma unroll
        for (uint32_t p = 0; p < n_preds; p++) {
            const PredD &pr = params.preds[p];
            if ((int)pr.proj != last_proj) {
                col_multi<R>(data, scratch, cols[pr.proj], row, q);
                last_proj = (int)pr.proj;
            }
            #pragma unroll
            for (int k = 0; k < R; k++)
                gv[k] = gv[k] | pred_eval(pr, q.v[k], q.fget(k));
            if (pr.gend) {
                #pragma unroll
                for (int k = 0; k < R; k++) { pv[k] = pv[k] & gv[k]; gv[k] = false; }
                /* early wave exit: if every lane's window is dead after
                 * this AND-group, skip the remaining pred columns' loads
                 * and evals (wave-uniform ballot branch) */
                bool alive = false;
                #pragma unroll
                for (int k = 0; k < R; k++) alive |= pv[k];
                if (__ballot(alive) == 0) break;
            }
        }
        bool any = false;
        #pragma unroll
        for (int k = 0; k < R; k++) any |= pv[k];
        if (__ballot(any) == 0) continue;
        #pragma unroll
        for (uint32_t a = 0; a < n_aggs; a++)
            acc_multi<R>(acc[a], params.aggs[a], data, scratch, cols, row, pv);
    }

    __shared__ ThreadAcc lds[AGG_BLOCK / WAVE][MAX_AGGS];
    const uint32_t wid = threadIdx.x / WAVE;
    const uint32_t lane = threadIdx.x % WAVE;
    #pragma unroll
    for (uint32_t a = 0; a < n_aggs; a++) {
        wave_reduce(acc[a], params.aggs[a].kind);
        if (lane == 0) lds[wid][a] = acc[a];
    }
    __syncthreads();
    if (wid == 0) {
        for (uint32_t a = lane; a < n_aggs; a += WAVE) {
            ThreadAcc r = lds[0][a];
            for (uint32_t w = 1; w < AGG_BLOCK / WAVE; w++)
                acc_merge(r, lds[w][a], params.aggs[a].kind);
            AccCell c;
            c.lo = r.lo; c.hi = r.hi; c.f = r.f; c.cnt = r.cnt;
            block_out[(uint64_t)blockIdx.x * n_aggs + a] = c;
        }
    }
}

/* =====================================================================
 * LDS-staged scan (experimental variant): the R-row kernel is latency
 * bound (SQ_WAIT_ANY ~72% of wave cycles even at 8 waves/SIMD). Here each
 * block BULK-LOADS its tile's canonical predicate streams into LDS with
 * coalesced dwordx4 loads (deep MLP, few waits), then rows extract values
 * at LDS latency. Aggregate operand columns stay in global memory behind
 * the wave-uniform ballot (rarely touched at Q6 selectivity).
 * Eligible when every projected column is dense canonical P/CONST.
 * ===================================================================== */

#define LDSK_TILE 2048

template <int NPREDS, int NAGGS>
__global__ __launch_bounds__(AGG_BLOCK) void lds_agg_kernel(
    const uint8_t *__restrict__ data, const uint8_t *__restrict__ scratch,
    const GroupDesc *__restrict__ groups, const ColLoc *__restrict__ colloc,
    AccCell *__restrict__ block_out, uint32_t tiles2_per_group,
    const AggParams params)
{
    const uint32_t n_preds = NPREDS >= 0 ? (uint32_t)NPREDS : params.n_preds;
    const uint32_t n_aggs = NAGGS >= 0 ? (uint32_t)NAGGS : params.n_aggs;
    const uint32_t gid = blockIdx.x / tiles2_per_group;
    const uint32_t tile = blockIdx.x % tiles2_per_group;
    const GroupDesc g = groups[gid];
    const ColLoc *cols = colloc + g.colbase;

    const uint32_t row_start = tile * LDSK_TILE;
    if (row_start >= g.row_count) {     /* empty tile: zero partials */
        if (threadIdx.x == 0)
            for (uint32_t a = 0; a < n_aggs; a++) {
                AccCell c{0, 0, 0.0, 0};
                ThreadAcc z;
                acc_init(z, params.aggs[a].kind);
                c.lo = z.lo; c.hi = z.hi; c.f = z.f; c.cnt = 0;
                block_out[(uint64_t)blockIdx.x * n_aggs + a] = c;
            }
        return;
    }
    const uint32_t row_end = min(row_start + LDSK_TILE, g.row_count);

    extern __shared__ uint8_t sld[];

    /* distinct predicate columns (wave-uniform bookkeeping) */
    uint32_t pcols[4];
    uint32_t npc = 0;
    for (uint32_t p = 0; p < n_preds; p++) {
        const uint32_t pj = params.preds[p].proj;
        bool seen = false;
        for (uint32_t i = 0; i < npc; i++) seen |= pcols[i] == pj;
        if (!seen && npc < 4) pcols[npc++] = pj;
    }

    /* phase A: stage each pred column's stream window (uniform control) */
    uint32_t roff[4], rdelta[4], rstep[4], rstart_pos[4];
    uint64_t rmask[4], rhval[4];
    uint8_t rmode[4];
    uint32_t lds_used = 0;
    for (uint32_t i = 0; i < npc; i++) {
        const ColLoc cl = cols[pcols[i]];
        rmode[i] = cl.mode;
        rhval[i] = (uint64_t)cl.hval;
        if (cl.mode == CSF_SEGMODE_CONST) {
            roff[i] = 0; rdelta[i] = 0; rstep[i] = 0; rstart_pos[i] = 0;
            rmask[i] = 0;
            continue;
        }
        const uint32_t Lx = cl.L, step = Lx + 3u;
        rstep[i] = step;
        rmask[i] = (~0ull) >> ((8u - Lx) * 8u);
        const uint32_t p0 = tile == 0 ? 0u
                            : row_start * step + (6u - Lx);
        rstart_pos[i] = p0;
        const uint64_t src0 = cl.val_off + p0;
        const uint64_t asrc = src0 & ~15ull;
        rdelta[i] = (uint32_t)(src0 - asrc);
        const uint32_t last_pos = (row_end - 1) * step + (6u - Lx);
        const uint32_t nbytes = (uint32_t)(cl.val_off + last_pos + 8 - asrc + 15) & ~15u;
        roff[i] = lds_used;
        /* cooperative 16B copy */
        const uint4 *gsrc = (const uint4 *)(data + asrc);
        uint4 *ldst = (uint4 *)(sld + lds_used);
        for (uint32_t o = threadIdx.x; o < (nbytes >> 4); o += AGG_BLOCK)
            ldst[o] = gsrc[o];
        lds_used += nbytes;
    }
    __syncthreads();

    ThreadAcc acc[NAGGS >= 0 ? NAGGS : MAX_AGGS];
    #pragma unroll
    for (uint32_t a = 0; a < n_aggs; a++) acc_init(acc[a], params.aggs[a].kind);

    /* extraction from LDS: value j of pred-col slot i */
    auto lval = [&](uint32_t i, uint32_t j) -> int64_t {
        if (rmode[i] == CSF_SEGMODE_CONST) return (int64_t)rhval[i];
        uint32_t pos = j * rstep[i] + (9u - rstep[i]);
        pos = j == 0 ? 1u : (j == 1 ? 9u : pos);
        const uint32_t rel = rdelta[i] + pos - rstart_pos[i];
        const uint32_t *l32 = (const uint32_t *)(sld + roff[i]);
        const uint32_t wi = rel >> 2, sh = (rel & 3u) * 8u;
        uint64_t lo = ((uint64_t)l32[wi + 1] << 32) | l32[wi];
        if (sh) lo = (lo >> sh) | ((uint64_t)l32[wi + 2] << (64u - sh));
        return (int64_t)((lo & rmask[i]) | rhval[i]);
    };

    constexpr int R = LDSK_TILE / AGG_BLOCK;      /* 8 rows per thread */
    const uint32_t my0 = row_start + threadIdx.x * R;
    bool pv[R], gv[R];
    #pragma unroll
    for (int k = 0; k < R; k++) {
        pv[k] = my0 + k < row_end;
        gv[k] = false;
    }
    int64_t v[R];
    int last_proj = -1;
    if (my0 < row_end) {
        #pragma unroll
        for (uint32_t p = 0; p < n_preds; p++) {
            const PredD &pr = params.preds[p];
            if ((int)pr.proj != last_proj) {
                uint32_t slot = 0;
                for (uint32_t i = 0; i < npc; i++)
                    if (pcols[i] == pr.proj) slot = i;
                #pragma unroll
                for (int k = 0; k < R; k++) v[k] = lval(slot, my0 + k);
                last_proj = (int)pr.proj;
            }
            #pragma unroll
            for (int k = 0; k < R; k++)
                gv[k] = gv[k] | pred_eval(pr, v[k], 0.0);   /* canon P: ints */
            if (pr.gend) {
                #pragma unroll
                for (int k = 0; k < R; k++) { pv[k] = pv[k] & gv[k]; gv[k] = false; }
            }
        }
    } else {
        #pragma unroll
        for (int k = 0; k < R; k++) pv[k] = false;
    }
    bool any = false;
    #pragma unroll
    for (int k = 0; k < R; k++) any |= pv[k];
    if (__ballot(any) != 0) {
        /* lanes past row_end load a safe in-bounds base; pv masks them */
        const uint32_t safe0 = my0 < row_end ? my0 : row_start;
        #pragma unroll
        for (uint32_t a = 0; a < n_aggs; a++)
            acc_multi<R>(acc[a], params.aggs[a], data, scratch, cols, safe0, pv);
    }

    __shared__ ThreadAcc lred2[AGG_BLOCK / WAVE][MAX_AGGS];
    const uint32_t wid = threadIdx.x / WAVE;
    const uint32_t lane = threadIdx.x % WAVE;
    #pragma unroll
    for (uint32_t a = 0; a < n_aggs; a++) {
        wave_reduce(acc[a], params.aggs[a].kind);
        if (lane == 0) lred2[wid][a] = acc[a];
    }
    __syncthreads();
    if (wid == 0) {
        for (uint32_t a = lane; a < n_aggs; a += WAVE) {
            ThreadAcc r = lred2[0][a];
            for (uint32_t w = 1; w < AGG_BLOCK / WAVE; w++)
                acc_merge(r, lred2[w][a], params.aggs[a].kind);
            AccCell c;
            c.lo = r.lo; c.hi = r.hi; c.f = r.f; c.cnt = r.cnt;
            block_out[(uint64_t)blockIdx.x * n_aggs + a] = c;
        }
    }
}

/* per-row precomputed aggregate contribution: operands loaded ONCE per row,
 * so the per-distinct-key reduce rounds touch registers only */
struct PrepAcc {
    int64_t lo, hi;    /* i128 contribution / i64 value */
    double f;
    bool valid;
};

__device__ inline void acc_prepare(PrepAcc &o, const AggD &g,
                                   const uint8_t *__restrict__ data,
                                   const uint8_t *__restrict__ scratch,
                                   const uint32_t *__restrict__ rank,
                                   const ColLoc *__restrict__ cols, uint32_t row)
{
    o.lo = 0; o.hi = 0; o.f = 0.0; o.valid = false;
    int64_t iv, ib, ic; double fv, fb, fc;
    switch (g.kind) {
        case CSTRIPE_AGG_COUNT_STAR:
            o.valid = true; o.lo = 1; break;
        case CSTRIPE_AGG_COUNT_COL:
            if (col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv)) { o.valid = true; o.lo = 1; }
            break;
        case CSTRIPE_AGG_SUM_I64:
        case CSTRIPE_AGG_MIN_I64:
        case CSTRIPE_AGG_MAX_I64:
            if (col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv)) {
                o.valid = true; o.lo = iv; o.hi = iv < 0 ? -1 : 0;
            }
            break;
        case CSTRIPE_AGG_SUM_F64:
        case CSTRIPE_AGG_MIN_F64:
        case CSTRIPE_AGG_MAX_F64:
            if (col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv)) { o.valid = true; o.f = fv; }
            break;
        case CSTRIPE_AGG_SUM_PROD_I64:
            if (col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv) &&
                col_value(data, scratch, rank, cols[g.proj_b], row, ib, fb)) {
                __int128 x = (__int128)iv * ib;
                o.valid = true; o.lo = (int64_t)(uint64_t)x; o.hi = (int64_t)(x >> 64);
            }
            break;
        case CSTRIPE_AGG_SUM_DISC_I64:
            if (col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv) &&
                col_value(data, scratch, rank, cols[g.proj_b], row, ib, fb)) {
                __int128 x = (__int128)iv * (g.one - ib);
                o.valid = true; o.lo = (int64_t)(uint64_t)x; o.hi = (int64_t)(x >> 64);
            }
            break;
        case CSTRIPE_AGG_SUM_DISC_TAX_I64:
            if (col_value(data, scratch, rank, cols[g.proj_a], row, iv, fv) &&
                col_value(data, scratch, rank, cols[g.proj_b], row, ib, fb) &&
                col_value(data, scratch, rank, cols[g.proj_c], row, ic, fc)) {
                __int128 x = (__int128)iv * (g.one - ib) * (g.one + ic);
                o.valid = true; o.lo = (int64_t)(uint64_t)x; o.hi = (int64_t)(x >> 64);
            }
            break;
    }
}


/* =====================================================================
 * Grouped aggregation (TPC-H Q1 shape): GROUP BY 1-2 categorical i8
 * columns, few distinct groups. The reference plan is worker HashAggregate
 * over ColumnarScan + coordinator merge per group (multi_explain.out:80-82,
 * multi_logical_optimizer.c:1807-1885). Here: per-WAVE LDS accumulator
 * tables (no atomics); rows reduce wave-cooperatively one distinct key at
 * a time (ballot + predicated identity + shuffle reduce); grid-stride
 * blocks flush one compact table each; a final single-block kernel merges.
 * ===================================================================== */

#define GRP_SLOTS CSTRIPE_MAX_GROUPS  /* distinct keys per wave table (a wave may see every group) */
#define GRP_GRID  1024        /* grid-stride blocks */

struct GroupParams {
    AggParams base;
    uint32_t n_group_cols;
    uint32_t gproj[CSTRIPE_MAX_GROUP_COLS];   /* projected slots of key cols */
    uint32_t n_work;                          /* n_groups * tiles_per_group */
    /* multi_grouped fused key decode: key cols stored as uniform 256 B
     * greedy-LZ4 segments are decoded into LDS inside the kernel (half-tile
     * granularity) instead of a scratch round trip */
    uint32_t kdec_mask;                       /* bit i: key col i from LDS */
    uint32_t kwidth[CSTRIPE_MAX_GROUP_COLS];  /* value width of key col i */
};


/* fold a prepared contribution into an LDS accumulator cell with atomics —
 * per-lane, no wave-cooperative rounds. int128 sums stay EXACT: the carry of
 * my own low-half add is detected from atomicAdd's return value and added to
 * the high half (independent of interleaving). f64 sums become
 * atomic-order-dependent (covered by the 1e-6 float tolerance). */
__device__ inline void acc_apply_atomic(ThreadAcc *cell, uint8_t kind, const PrepAcc &p)
{
    if (!p.valid) return;
    switch (kind) {
        case CSTRIPE_AGG_MIN_I64:
            atomicMin((long long *)&cell->lo, (long long)p.lo);
            break;
        case CSTRIPE_AGG_MAX_I64:
            atomicMax((long long *)&cell->lo, (long long)p.lo);
            break;
        case CSTRIPE_AGG_MIN_F64: {
            double v = p.f;
            unsigned long long *addr = (unsigned long long *)&cell->f;
            unsigned long long cur = *addr;
            while (true) {
                double c;
                memcpy(&c, &cur, 8);
                if (f64cmp_pg(v, c) >= 0) break;   /* PG order (NaN high) */
                unsigned long long nv;
                memcpy(&nv, &v, 8);
                unsigned long long prev = atomicCAS(addr, cur, nv);
                if (prev == cur) break;
                cur = prev;
            }
            break;
        }
        case CSTRIPE_AGG_MAX_F64: {
            double v = p.f;
            unsigned long long *addr = (unsigned long long *)&cell->f;
            unsigned long long cur = *addr;
            while (true) {
                double c;
                memcpy(&c, &cur, 8);
                if (f64cmp_pg(v, c) <= 0) break;   /* PG order (NaN high) */
                unsigned long long nv;
                memcpy(&nv, &v, 8);
                unsigned long long prev = atomicCAS(addr, cur, nv);
                if (prev == cur) break;
                cur = prev;
            }
            break;
        }
        case CSTRIPE_AGG_SUM_F64:
            atomicAdd((double *)&cell->f, p.f);
            break;
        default: {   /* counts + i128 sums */
            unsigned long long old =
                atomicAdd((unsigned long long *)&cell->lo, (unsigned long long)p.lo);
            int64_t carry = (old + (unsigned long long)p.lo) < old ? 1 : 0;
            int64_t hi = p.hi + carry;
            if (hi) atomicAdd((unsigned long long *)&cell->hi, (unsigned long long)hi);
            break;
        }
    }
    atomicAdd((unsigned long long *)&cell->cnt, 1ull);
}

/* =====================================================================
 * Multi-row GROUPED kernel for ALL-DENSE chunks (TPC-H Q1 shape): the same
 * R-row windowed loads as multi_agg_kernel (the scan is load-latency
 * bound), folding into a per-WAVE 16-slot LDS group table with the
 * carry-exact atomics. >16 distinct keys set device flag 8 and the host
 * re-runs the 64-group fallback (grouped_agg_kernel).
 * ===================================================================== */

#define MGRP_SLOTS 16

/* value-only LDS fold: multi_grouped tracks ONE per-slot row counter (all
 * operands are dense there, so every agg's count equals the group's row
 * count) — halves the per-row LDS atomic traffic vs acc_apply_atomic */
__device__ inline void acc_apply_value_atomic(ThreadAcc *cell, uint8_t kind,
                                              const PrepAcc &p)
{
    switch (kind) {
        case CSTRIPE_AGG_COUNT_STAR:
        case CSTRIPE_AGG_COUNT_COL:
            return;                         /* derived from the slot counter */
        case CSTRIPE_AGG_MIN_I64:
            atomicMin((long long *)&cell->lo, (long long)p.lo);
            return;
        case CSTRIPE_AGG_MAX_I64:
            atomicMax((long long *)&cell->lo, (long long)p.lo);
            return;
        case CSTRIPE_AGG_SUM_F64:
            atomicAdd((double *)&cell->f, p.f);
            return;
        case CSTRIPE_AGG_MIN_F64:
        case CSTRIPE_AGG_MAX_F64: {
            unsigned long long *addr = (unsigned long long *)&cell->f;
            unsigned long long cur = *addr;
            const bool want_min = kind == CSTRIPE_AGG_MIN_F64;
            while (true) {
                double c;
                memcpy(&c, &cur, 8);
                const int cmp = f64cmp_pg(p.f, c);
                if (want_min ? cmp >= 0 : cmp <= 0) break;
                unsigned long long nv;
                memcpy(&nv, &p.f, 8);
                unsigned long long prev = atomicCAS(addr, cur, nv);
                if (prev == cur) break;
                cur = prev;
            }
            return;
        }
        default: {                          /* i128 sums, carry-exact */
            unsigned long long old =
                atomicAdd((unsigned long long *)&cell->lo, (unsigned long long)p.lo);
            int64_t carry = (old + (unsigned long long)p.lo) < old ? 1 : 0;
            int64_t hi = p.hi + carry;
            if (hi) atomicAdd((unsigned long long *)&cell->hi, (unsigned long long)hi);
            return;
        }
    }
}

template <int NAGGS, int R, int MINW = 4>
__global__ __launch_bounds__(AGG_BLOCK, MINW) void multi_grouped_kernel(
    const uint8_t *__restrict__ data, const uint8_t *__restrict__ scratch,
    const GroupDesc *__restrict__ groups, const ColLoc *__restrict__ colloc,
    const SegDesc *__restrict__ segs, const uint32_t *__restrict__ segstart,
    uint32_t *__restrict__ keys_out, AccCell *__restrict__ cells_out,
    int *__restrict__ err, const GroupParams gp)
{
    const AggParams &params = gp.base;
    const uint32_t n_aggs = NAGGS >= 0 ? (uint32_t)NAGGS : params.n_aggs;
    const uint32_t wid = threadIdx.x / WAVE;
    const uint32_t lane = threadIdx.x % WAVE;
    const uint32_t n_waves = AGG_BLOCK / WAVE;
    constexpr uint32_t HALF = (uint32_t)R * AGG_BLOCK;   /* rows per pass */

    __shared__ int wkeys[AGG_BLOCK / WAVE][MGRP_SLOTS];
    __shared__ unsigned long long wcnt[AGG_BLOCK / WAVE][MGRP_SLOTS];
    extern __shared__ uint8_t mg_lds[];
    ThreadAcc *wacc = (ThreadAcc *)mg_lds;   /* [wave][slot][agg] */
    /* fused key-decode regions follow the accumulator tables: per decoded
     * key col, ceil(HALF*width/256) lanes x 280 B (the lane-decode stride) */
    uint32_t kbase[CSTRIPE_MAX_GROUP_COLS] = {0, 0};
    uint32_t ksegs_half[CSTRIPE_MAX_GROUP_COLS] = {0, 0};
    {
        uint32_t off = (uint32_t)(n_waves * MGRP_SLOTS * n_aggs * sizeof(ThreadAcc));
        off = (off + 15u) & ~15u;
        for (uint32_t i = 0; i < gp.n_group_cols; i++) {
            if (!(gp.kdec_mask & (1u << i))) continue;
            kbase[i] = off;
            ksegs_half[i] = (HALF * gp.kwidth[i] + 255u) / 256u;
            off += ksegs_half[i] * 280u;
        }
    }

    for (uint32_t i = threadIdx.x; i < n_waves * MGRP_SLOTS; i += AGG_BLOCK) {
        wkeys[i / MGRP_SLOTS][i % MGRP_SLOTS] = -1;
        wcnt[i / MGRP_SLOTS][i % MGRP_SLOTS] = 0;
        for (uint32_t a = 0; a < n_aggs; a++)
            acc_init(wacc[i * n_aggs + a], params.aggs[a].kind);
    }
    __syncthreads();

    ThreadAcc *myacc = wacc + (size_t)wid * MGRP_SLOTS * n_aggs;
    int *mykeys = wkeys[wid];
    unsigned long long *mycnt = wcnt[wid];
    uint32_t ck0 = ~0u, ck1 = ~0u, cs0 = 0, cs1 = 0;   /* key->slot cache */

    /* grid-stride over (chunk, tile) work items: tables persist across
     * items and flush once, keeping per-block output buffers small */
    for (uint32_t work = blockIdx.x; work < gp.n_work; work += gridDim.x) {
    const uint32_t gid = work / params.tiles_per_group;
    const uint32_t tile = work % params.tiles_per_group;
    const GroupDesc g = groups[gid];
    const ColLoc *cols = colloc + g.colbase;
    const uint32_t row_start = tile * TILE_ROWS;
    const uint32_t row_end = min(row_start + TILE_ROWS, g.row_count);

    for (uint32_t half_start = row_start; half_start < row_end;
         half_start += HALF) {
        const uint32_t half_end = min(half_start + HALF, row_end);

        /* phase 1: decode this half's key-column segments into LDS (uniform
         * 256 B greedy-LZ4 segments, one lane each) */
        if (gp.kdec_mask) {
            __syncthreads();           /* LDS regions reused across halves */
            for (uint32_t i = 0; i < gp.n_group_cols; i++) {
                if (!(gp.kdec_mask & (1u << i))) continue;
                const uint32_t base_lane = i * 64u;
                const uint32_t rows_half = half_end - half_start;
                const uint32_t nseg = (rows_half * gp.kwidth[i] + 255u) / 256u;
                const uint32_t k = threadIdx.x - base_lane;
                if (threadIdx.x >= base_lane && k < nseg) {
                    const uint32_t pj = gp.gproj[i];
                    const uint32_t seg0 = segstart[(uint64_t)gid * params.n_proj + pj]
                                          + (half_start * gp.kwidth[i]) / 256u;
                    const SegDesc sd = segs[seg0 + k];
                    lz4_lane_decode(data, sd,
                                    mg_lds + kbase[i] + (size_t)k * 280u, err);
                }
            }
            __syncthreads();
        }

        /* NOTE: no `continue` below this point — every thread must reach
         * the next half's __syncthreads (the key-decode LDS is reused) */
        const uint32_t my0 = half_start + R * threadIdx.x;
        bool pv[R], gv[R];
        #pragma unroll
        for (int k = 0; k < R; k++) { pv[k] = my0 + k < half_end; gv[k] = false; }
        int last_proj = -1;
        ValsR<R> q{};
        #pragma unroll
        for (uint32_t pp = 0; pp < params.n_preds; pp++) {
            const PredD &pr = params.preds[pp];
            if ((int)pr.proj != last_proj) {
                col_multi<R>(data, scratch, cols[pr.proj],
                             my0 < half_end ? my0 : half_start, q);
                last_proj = (int)pr.proj;
            }
            #pragma unroll
            for (int k = 0; k < R; k++)
                gv[k] = gv[k] | pred_eval(pr, q.v[k], q.fget(k));
            if (pr.gend) {
                #pragma unroll
                for (int k = 0; k < R; k++) { pv[k] = pv[k] & gv[k]; gv[k] = false; }
                /* early wave exit after a dead AND-group (see multi_agg) */
                bool alive = false;
                #pragma unroll
                for (int k = 0; k < R; k++) alive |= pv[k];
                if (__ballot(alive) == 0) break;
            }
        }
        bool any = false;
        #pragma unroll
        for (int k = 0; k < R; k++) any |= pv[k];
        if (__ballot(any) != 0) {

        /* group keys: LDS-decoded cols read at half-local offsets, others
         * through col_multi; 9-bit enc (bit 8 = NULL never set: dense) */
        uint32_t key[R];
        {
            const uint32_t safe0 = my0 < half_end ? my0 : half_start;
            #pragma unroll
            for (int k = 0; k < R; k++) key[k] = 0;
            for (uint32_t i = 0; i < gp.n_group_cols; i++) {
                const uint32_t pj = gp.gproj[i];
                const bool is_text = cols[pj].type == CSTRIPE_TEXT;
                if (gp.kdec_mask & (1u << i)) {
                    const uint32_t kw = gp.kwidth[i];
                    #pragma unroll
                    for (int k = 0; k < R; k++) {
                        const uint32_t r = (safe0 + k) - half_start;
                        const uint32_t bo = r * kw;
                        const uint8_t *pp2 = mg_lds + kbase[i]
                                             + (bo >> 8) * 280u + (bo & 255u);
                        uint32_t raw = kw == 4 ? *(const uint32_t *)pp2
                                               : (uint32_t)*pp2;
                        if (is_text) raw >>= 8;
                        key[k] |= (raw & 0xFF) << (9 * i);
                    }
                } else {
                    ValsR<R> kq;
                    col_multi<R>(data, scratch, cols[pj], safe0, kq);
                    #pragma unroll
                    for (int k = 0; k < R; k++) {
                        uint32_t raw = (uint32_t)kq.v[k];
                        if (is_text) raw >>= 8;
                        key[k] |= (raw & 0xFF) << (9 * i);
                    }
                }
            }
        }
        /* slot per row: register cache, else LDS claim (CAS linear probe) */
        uint32_t slot[R];
        #pragma unroll
        for (int k = 0; k < R; k++) {
            slot[k] = 0xFFFFFFFF;
            if (!pv[k]) continue;
            if (key[k] == ck0) slot[k] = cs0;
            else if (key[k] == ck1) slot[k] = cs1;
            else {
                for (uint32_t s2 = 0; s2 < MGRP_SLOTS; s2++) {
                    int old = atomicCAS(&mykeys[s2], -1, (int)key[k]);
                    if (old == -1 || old == (int)key[k]) { slot[k] = s2; break; }
                }
                if (slot[k] == 0xFFFFFFFF) { atomicOr(err, 8); pv[k] = false; continue; }
                ck1 = ck0; cs1 = cs0;
                ck0 = key[k]; cs0 = slot[k];
            }
        }

        /* one row counter per slot (dense: every agg's count equals it) */
        #pragma unroll
        for (int k = 0; k < R; k++)
            if (pv[k]) atomicAdd(&mycnt[slot[k]], 1ull);

        /* operand loads once per agg column, contributions folded per row */
        const uint32_t safe0 = my0 < half_end ? my0 : half_start;
        #pragma unroll
        for (uint32_t a = 0; a < n_aggs; a++) {
            const AggD &ag = params.aggs[a];
            const uint8_t kind = ag.kind;
            ValsR<R> qa, qb, qc;
            if (kind != CSTRIPE_AGG_COUNT_STAR && kind != CSTRIPE_AGG_COUNT_COL) {
                col_multi<R>(data, scratch, cols[ag.proj_a], safe0, qa);
                if (kind >= CSTRIPE_AGG_SUM_PROD_I64)
                    col_multi<R>(data, scratch, cols[ag.proj_b], safe0, qb);
                if (kind == CSTRIPE_AGG_SUM_DISC_TAX_I64)
                    col_multi<R>(data, scratch, cols[ag.proj_c], safe0, qc);
            }
            #pragma unroll
            for (int k = 0; k < R; k++) {
                if (!pv[k]) continue;
                PrepAcc pc;
                pc.valid = true; pc.f = 0.0; pc.lo = 0; pc.hi = 0;
                switch (kind) {
                    case CSTRIPE_AGG_COUNT_STAR:
                    case CSTRIPE_AGG_COUNT_COL: pc.lo = 1; break;
                    case CSTRIPE_AGG_SUM_I64:
                    case CSTRIPE_AGG_MIN_I64:
                    case CSTRIPE_AGG_MAX_I64:
                        pc.lo = qa.v[k]; pc.hi = qa.v[k] < 0 ? -1 : 0; break;
                    case CSTRIPE_AGG_SUM_F64:
                    case CSTRIPE_AGG_MIN_F64:
                    case CSTRIPE_AGG_MAX_F64:
                        pc.f = qa.fget(k); break;
                    case CSTRIPE_AGG_SUM_PROD_I64: {
                        __int128 x = (__int128)qa.v[k] * qb.v[k];
                        pc.lo = (int64_t)(uint64_t)x; pc.hi = (int64_t)(x >> 64); break;
                    }
                    case CSTRIPE_AGG_SUM_DISC_I64: {
                        __int128 x = (__int128)qa.v[k] * (ag.one - qb.v[k]);
                        pc.lo = (int64_t)(uint64_t)x; pc.hi = (int64_t)(x >> 64); break;
                    }
                    case CSTRIPE_AGG_SUM_DISC_TAX_I64: {
                        __int128 x = (__int128)qa.v[k] * (ag.one - qb.v[k])
                                     * (ag.one + qc.v[k]);
                        pc.lo = (int64_t)(uint64_t)x; pc.hi = (int64_t)(x >> 64); break;
                    }
                    default: pc.valid = false; break;
                }
                acc_apply_value_atomic(&myacc[slot[k] * n_aggs + a], kind, pc);
            }
        }
        }   /* wave has passing rows */
    }   /* half loop */
    }   /* work loop */
    __syncthreads();

    /* merge the block's wave tables into one compact list (as grouped_agg) */
    if (threadIdx.x == 0) {
        uint32_t *bk = keys_out + (size_t)blockIdx.x * (n_waves * MGRP_SLOTS);
        AccCell *bc = cells_out + (size_t)blockIdx.x * (n_waves * MGRP_SLOTS) * n_aggs;
        uint32_t n = 0;
        for (uint32_t w = 0; w < n_waves; w++) {
            for (uint32_t s2 = 0; s2 < MGRP_SLOTS; s2++) {
                int kk = wkeys[w][s2];
                if (kk < 0) continue;
                uint32_t at = n;
                for (uint32_t j = 0; j < n; j++) if (bk[j] == (uint32_t)kk) { at = j; break; }
                ThreadAcc *src = wacc + ((size_t)(w * MGRP_SLOTS) + s2) * n_aggs;
                const int64_t rowcnt = (int64_t)wcnt[w][s2];
                for (uint32_t a = 0; a < n_aggs; a++) {
                    src[a].cnt = rowcnt;
                    const uint8_t kd = params.aggs[a].kind;
                    if (kd == CSTRIPE_AGG_COUNT_STAR || kd == CSTRIPE_AGG_COUNT_COL)
                        src[a].lo = rowcnt;
                }
                if (at == n) {
                    bk[n] = (uint32_t)kk;
                    for (uint32_t a = 0; a < n_aggs; a++) {
                        AccCell c{src[a].lo, src[a].hi, src[a].f, src[a].cnt};
                        bc[(size_t)n * n_aggs + a] = c;
                    }
                    n++;
                } else {
                    for (uint32_t a = 0; a < n_aggs; a++) {
                        AccCell c = bc[(size_t)at * n_aggs + a];
                        ThreadAcc cur{c.lo, c.hi, c.f, c.cnt};
                        acc_merge(cur, src[a], params.aggs[a].kind);
                        AccCell o{cur.lo, cur.hi, cur.f, cur.cnt};
                        bc[(size_t)at * n_aggs + a] = o;
                    }
                }
            }
        }
        for (uint32_t j = n; j < n_waves * MGRP_SLOTS; j++) bk[j] = 0xFFFFFFFFu;
    }
}

template <int NAGGS>
__global__ __launch_bounds__(AGG_BLOCK) void grouped_agg_kernel(
    const uint8_t *__restrict__ data, const uint8_t *__restrict__ scratch,
    const uint32_t *__restrict__ rank, const GroupDesc *__restrict__ groups,
    const ColLoc *__restrict__ colloc, uint32_t *__restrict__ keys_out,
    AccCell *__restrict__ cells_out, int *__restrict__ err,
    const GroupParams gp)
{
    const AggParams &params = gp.base;
    /* compile-time agg count unrolls the prep/reduce loops so the per-row
     * PrepAcc array stays in registers instead of scratch */
    const uint32_t n_aggs_ct = NAGGS >= 0 ? (uint32_t)NAGGS : params.n_aggs;
    const uint32_t wid = threadIdx.x / WAVE;
    const uint32_t lane = threadIdx.x % WAVE;
    const uint32_t n_waves = AGG_BLOCK / WAVE;

    extern __shared__ uint8_t gsh[];
    uint32_t *wkeys = (uint32_t *)gsh;                          /* [wave][S] */
    ThreadAcc *wacc = (ThreadAcc *)(gsh + n_waves * GRP_SLOTS * sizeof(uint32_t));
    /* wacc[((wave*S)+slot)*n_aggs + a] */

    for (uint32_t i = threadIdx.x; i < n_waves * GRP_SLOTS; i += AGG_BLOCK)
        wkeys[i] = 0xFFFFFFFFu;
    __syncthreads();

    uint32_t *mykeys = wkeys + wid * GRP_SLOTS;
    ThreadAcc *myacc = wacc + (size_t)wid * GRP_SLOTS * n_aggs_ct;
    uint32_t used = 0;       /* wave-uniform slot count (updated by lane 0 path) */
    /* per-lane 4-way key->slot register cache: hits skip the wave-ballot
     * slot rounds entirely (few distinct groups => near-100% after warmup) */
    uint32_t ck0 = ~0u, ck1 = ~0u, ck2 = ~0u, ck3 = ~0u;
    uint32_t cs0 = 0, cs1 = 0, cs2 = 0, cs3 = 0;

    for (uint32_t work = blockIdx.x; work < gp.n_work; work += gridDim.x) {
        const uint32_t gid = work / params.tiles_per_group;
        const uint32_t tile = work % params.tiles_per_group;
        const GroupDesc g = groups[gid];
        const ColLoc *cols = colloc + g.colbase;
        uint32_t row_start = tile * TILE_ROWS;
        uint32_t row_end = min(row_start + TILE_ROWS, g.row_count);

        for (uint32_t base_row = row_start + wid * WAVE; base_row < row_end;
             base_row += WAVE * n_waves) {
            uint32_t row = base_row + lane;
            bool pass = row < row_end;
            if (pass) {
                int last_proj = -1;
                int64_t liv = 0; double lfv = 0; bool lok = false;
                bool gv = false;
                for (uint32_t p = 0; p < params.n_preds; p++) {
                    const PredD &pr = params.preds[p];
                    if ((int)pr.proj != last_proj) {
                        lok = col_value(data, scratch, rank, cols[pr.proj], row, liv, lfv);
                        last_proj = (int)pr.proj;
                    }
                    gv = gv | (lok && pred_eval(pr, liv, lfv));
                    if (pr.gend) { pass = pass & gv; gv = false; }
                }
            }
            uint32_t key = 0;
            if (pass) {
                int64_t kv; double kf;
                for (uint32_t gc = 0; gc < gp.n_group_cols; gc++) {
                    /* NULL keys form their own group (reference
                     * HashAggregate treats NULLs as equal): 9-bit encoding,
                     * bit 8 = null. TEXT keys group by the first payload
                     * byte of the varlena slot (char(1) semantics). */
                    const ColLoc &kc = cols[gp.gproj[gc]];
                    bool ok = col_value(data, scratch, rank, kc, row, kv, kf);
                    uint32_t raw = (uint32_t)kv;
                    if (kc.type == CSTRIPE_TEXT) raw >>= 8;
                    uint32_t enc = ok ? (raw & 0xFF) : CSTRIPE_GROUP_KEY_NULL;
                    key |= enc << (9 * gc);
                }
            }
            /* round 0 (agg-independent): assign each passing lane its key's
             * wave-table slot — register cache first, ballot rounds only
             * for misses */
            uint32_t myslot = 0xFFFFFFFF;
            if (pass) {
                if (key == ck0) myslot = cs0;
                else if (key == ck1) myslot = cs1;
                else if (key == ck2) myslot = cs2;
                else if (key == ck3) myslot = cs3;
            }
            {
                uint64_t remaining = __ballot(pass && myslot == 0xFFFFFFFF);
                while (remaining) {
                    int leader = __ffsll((unsigned long long)remaining) - 1;
                    uint32_t kk = (uint32_t)__shfl((int)key, leader, WAVE);
                    bool mine = pass && key == kk;
                    uint32_t slot = 0xFFFFFFFF;
                    for (uint32_t sidx = 0; sidx < used; sidx++)
                        if (mykeys[sidx] == kk) { slot = sidx; break; }
                    if (slot == 0xFFFFFFFF) {
                        if (used >= GRP_SLOTS) {      /* overflow: flag + drop */
                            if (lane == 0) atomicOr(err, 8);
                            remaining &= ~__ballot(mine);
                            continue;
                        }
                        slot = used++;
                        if (lane == 0) mykeys[slot] = kk;
                        for (uint32_t a = 0; a < n_aggs_ct; a++) {
                            ThreadAcc z;
                            acc_init(z, params.aggs[a].kind);
                            if (lane == 0) myacc[slot * n_aggs_ct + a] = z;
                        }
                    }
                    if (mine) {
                        myslot = slot;
                        ck3 = ck2; cs3 = cs2;
                        ck2 = ck1; cs2 = cs1;
                        ck1 = ck0; cs1 = cs0;
                        ck0 = key; cs0 = slot;
                    }
                    remaining &= ~__ballot(mine);
                }
            }
            /* per agg (unrolled): one prepared contribution per row folded
             * into the wave's LDS table with per-lane atomics */
            if (pass && myslot != 0xFFFFFFFF) {
                for (uint32_t a = 0; a < n_aggs_ct; a++) {
                    PrepAcc p;
                    acc_prepare(p, params.aggs[a], data, scratch, rank, cols, row);
                    acc_apply_atomic(&myacc[myslot * n_aggs_ct + a],
                                     params.aggs[a].kind, p);
                }
            }
        }
    }
    __syncthreads();

    /* merge the block's wave tables into one compact list; write to global.
     * wave 0 lane 0 does it serially — tables are tiny. */
    if (threadIdx.x == 0) {
        uint32_t *bk = keys_out + (size_t)blockIdx.x * (n_waves * GRP_SLOTS);
        AccCell *bc = cells_out + (size_t)blockIdx.x * (n_waves * GRP_SLOTS) * n_aggs_ct;
        uint32_t n = 0;
        for (uint32_t w = 0; w < n_waves; w++) {
            for (uint32_t sidx = 0; sidx < GRP_SLOTS; sidx++) {
                uint32_t k = wkeys[w * GRP_SLOTS + sidx];
                if (k == 0xFFFFFFFFu) continue;
                uint32_t at = n;
                for (uint32_t j = 0; j < n; j++) if (bk[j] == k) { at = j; break; }
                ThreadAcc *src = wacc + ((size_t)(w * GRP_SLOTS) + sidx) * n_aggs_ct;
                if (at == n) {
                    bk[n] = k;
                    for (uint32_t a = 0; a < n_aggs_ct; a++) {
                        AccCell c{src[a].lo, src[a].hi, src[a].f, src[a].cnt};
                        bc[(size_t)n * n_aggs_ct + a] = c;
                    }
                    n++;
                } else {
                    for (uint32_t a = 0; a < n_aggs_ct; a++) {
                        AccCell c = bc[(size_t)at * n_aggs_ct + a];
                        ThreadAcc cur{c.lo, c.hi, c.f, c.cnt};
                        acc_merge(cur, src[a], params.aggs[a].kind);
                        AccCell o{cur.lo, cur.hi, cur.f, cur.cnt};
                        bc[(size_t)at * n_aggs_ct + a] = o;
                    }
                }
            }
        }
        for (uint32_t j = n; j < n_waves * GRP_SLOTS; j++) bk[j] = 0xFFFFFFFFu;
    }
}

/* =====================================================================
 * FUSED GROUPED kernel (TPC-H Q1 shape): same no-scratch idea as
 * fused_agg_kernel, generalized to mixed i64/i8 columns — tile = 1536 rows
 * (1536*width is a multiple of 256 for both widths, so tile boundaries land
 * on segment boundaries), lane ranges per column by prefix. Groups
 * accumulate into ONE per-block 16-slot LDS table with the carry-exact
 * atomics; blocks grid-stride over tiles and flush their table once.
 * ===================================================================== */

#define FUSEG_TILE_ROWS 1536
#define FUSEG_SLOTS 16
#define FUSEG_GRID 2048

struct FusedTileG {
    uint32_t row_count;
    uint32_t seg_base[8];
};

struct FusedGParams {
    AggParams base;
    uint32_t n_group_cols;
    uint32_t gproj[CSTRIPE_MAX_GROUP_COLS];
    uint32_t n_tiles;
    uint16_t lane_base[8];       /* per proj col: first LDS region */
    uint8_t  width[8];           /* per proj col: value width (1 or 8) */
    uint16_t lanes_total;
};

template <int NAGGS>
__global__ __launch_bounds__(AGG_BLOCK) void fused_grouped_kernel(
    const uint8_t *__restrict__ data, const SegDesc *__restrict__ segs,
    const FusedTileG *__restrict__ tiles, uint32_t *__restrict__ keys_out,
    AccCell *__restrict__ cells_out, int *__restrict__ err,
    const FusedGParams gp)
{
    const AggParams &params = gp.base;
    const uint32_t n_aggs_ct = NAGGS >= 0 ? (uint32_t)NAGGS : params.n_aggs;
    const uint32_t tid = threadIdx.x;
    extern __shared__ uint8_t lds[];
    int *skeys = (int *)(lds + (size_t)gp.lanes_total * FUSE_STRIDE);
    ThreadAcc *scells = (ThreadAcc *)(skeys + FUSEG_SLOTS);

    for (uint32_t i = tid; i < FUSEG_SLOTS; i += AGG_BLOCK) {
        skeys[i] = -1;
        for (uint32_t a = 0; a < n_aggs_ct; a++)
            acc_init(scells[i * n_aggs_ct + a], params.aggs[a].kind);
    }
    __syncthreads();

    /* my (col, seg) assignment: first col whose lane range contains tid */
    uint32_t mycol = 0xFFFFFFFF, myseg = 0;
    for (uint32_t c2 = 0; c2 < params.n_proj; c2++) {
        uint32_t base = gp.lane_base[c2];
        uint32_t span = (uint32_t)FUSEG_TILE_ROWS * gp.width[c2] / 256;
        if (tid >= base && tid < base + span) { mycol = c2; myseg = tid - base; }
    }

    auto val = [&](uint32_t proj, uint32_t row) -> int64_t {
        const uint32_t byteoff = row * gp.width[proj];
        const uint8_t *p = lds + ((size_t)gp.lane_base[proj] + (byteoff >> 8)) * FUSE_STRIDE
                           + (byteoff & 255);
        return gp.width[proj] == 8 ? *(const int64_t *)p : (int64_t)*(const int8_t *)p;
    };

    /* per-lane register key->slot cache */
    uint32_t ck0 = ~0u, ck1 = ~0u;
    uint32_t cs0 = 0, cs1 = 0;

    for (uint32_t work = blockIdx.x; work < gp.n_tiles; work += gridDim.x) {
        const FusedTileG t = tiles[work];

        /* phase 1: decode */
        if (mycol != 0xFFFFFFFF) {
            const uint32_t nsegs = (t.row_count * gp.width[mycol] + 255) >> 8;
            if (myseg < nsegs) {
                const SegDesc sd = segs[t.seg_base[mycol] + myseg];
                lz4_lane_decode(data, sd, lds + (size_t)tid * FUSE_STRIDE, err);
            }
        }
        __syncthreads();

        /* phase 2: filter + group + accumulate */
        for (uint32_t row = tid; row < t.row_count; row += AGG_BLOCK) {
            bool pass = true, gv = false;
            int last_proj = -1;
            int64_t liv = 0;
            for (uint32_t p = 0; p < params.n_preds; p++) {
                const PredD &pr = params.preds[p];
                if ((int)pr.proj != last_proj) { liv = val(pr.proj, row); last_proj = (int)pr.proj; }
                gv = gv | pred_eval(pr, liv, 0.0);
                if (pr.gend) { pass = pass & gv; gv = false; }
            }
            if (!pass) continue;
            uint32_t key = 0;
            for (uint32_t gc = 0; gc < gp.n_group_cols; gc++)
                key |= ((uint32_t)val(gp.gproj[gc], row) & 0xFF) << (9 * gc);
            uint32_t slot = 0xFFFFFFFF;
            if (key == ck0) slot = cs0;
            else if (key == ck1) slot = cs1;
            else {
                for (uint32_t s2 = 0; s2 < FUSEG_SLOTS; s2++) {
                    int old = atomicCAS(&skeys[s2], -1, (int)key);
                    if (old == -1 || old == (int)key) { slot = s2; break; }
                }
                if (slot == 0xFFFFFFFF) { atomicOr(err, 8); continue; }
                ck1 = ck0; cs1 = cs0;
                ck0 = key; cs0 = slot;
            }
            for (uint32_t a = 0; a < n_aggs_ct; a++) {
                const AggD &g = params.aggs[a];
                PrepAcc p2;
                p2.valid = true; p2.f = 0.0; p2.hi = 0;
                switch (g.kind) {
                    case CSTRIPE_AGG_COUNT_STAR:
                    case CSTRIPE_AGG_COUNT_COL: p2.lo = 1; break;
                    case CSTRIPE_AGG_SUM_I64:
                    case CSTRIPE_AGG_MIN_I64:
                    case CSTRIPE_AGG_MAX_I64: {
                        int64_t v = val(g.proj_a, row);
                        p2.lo = v; p2.hi = v < 0 ? -1 : 0; break;
                    }
                    case CSTRIPE_AGG_SUM_PROD_I64: {
                        __int128 x = (__int128)val(g.proj_a, row) * val(g.proj_b, row);
                        p2.lo = (int64_t)(uint64_t)x; p2.hi = (int64_t)(x >> 64); break;
                    }
                    case CSTRIPE_AGG_SUM_DISC_I64: {
                        __int128 x = (__int128)val(g.proj_a, row) * (g.one - val(g.proj_b, row));
                        p2.lo = (int64_t)(uint64_t)x; p2.hi = (int64_t)(x >> 64); break;
                    }
                    case CSTRIPE_AGG_SUM_DISC_TAX_I64: {
                        __int128 x = (__int128)val(g.proj_a, row) * (g.one - val(g.proj_b, row))
                                     * (g.one + val(g.proj_c, row));
                        p2.lo = (int64_t)(uint64_t)x; p2.hi = (int64_t)(x >> 64); break;
                    }
                    default: p2.valid = false; break;
                }
                acc_apply_atomic(&scells[slot * n_aggs_ct + a], g.kind, p2);
            }
        }
        __syncthreads();     /* LDS decode regions reused next tile */
    }

    /* flush the block table */
    __syncthreads();
    if (tid == 0) {
        uint32_t *bk = keys_out + (size_t)blockIdx.x * FUSEG_SLOTS;
        AccCell *bc = cells_out + (size_t)blockIdx.x * FUSEG_SLOTS * n_aggs_ct;
        for (uint32_t s2 = 0; s2 < FUSEG_SLOTS; s2++) {
            bk[s2] = skeys[s2] < 0 ? 0xFFFFFFFFu : (uint32_t)skeys[s2];
            for (uint32_t a = 0; a < n_aggs_ct; a++) {
                const ThreadAcc &x = scells[s2 * n_aggs_ct + a];
                AccCell cell{x.lo, x.hi, x.f, x.cnt};
                bc[(size_t)s2 * n_aggs_ct + a] = cell;
            }
        }
    }
}

/* single-block merge of per-block group tables -> final <=64 groups.
 * Parallel: 256 threads stride the (block, slot) entries; keys claim final
 * slots via LDS CAS (linear probe); cells merge with the same atomic
 * carry-exact machinery as the main kernel. */
__device__ inline void cell_merge_atomic(ThreadAcc *dst, uint8_t kind, const AccCell &src)
{
    switch (kind) {
        case CSTRIPE_AGG_MIN_I64:
            atomicMin((long long *)&dst->lo, (long long)src.lo);
            break;
        case CSTRIPE_AGG_MAX_I64:
            atomicMax((long long *)&dst->lo, (long long)src.lo);
            break;
        case CSTRIPE_AGG_MIN_F64: {
            unsigned long long *addr = (unsigned long long *)&dst->f;
            unsigned long long cur = *addr;
            while (true) {
                double c;
                memcpy(&c, &cur, 8);
                if (f64cmp_pg(src.f, c) >= 0) break;   /* PG order (NaN high) */
                unsigned long long nv;
                memcpy(&nv, &src.f, 8);
                unsigned long long prev = atomicCAS(addr, cur, nv);
                if (prev == cur) break;
                cur = prev;
            }
            break;
        }
        case CSTRIPE_AGG_MAX_F64: {
            unsigned long long *addr = (unsigned long long *)&dst->f;
            unsigned long long cur = *addr;
            while (true) {
                double c;
                memcpy(&c, &cur, 8);
                if (f64cmp_pg(src.f, c) <= 0) break;   /* PG order (NaN high) */
                unsigned long long nv;
                memcpy(&nv, &src.f, 8);
                unsigned long long prev = atomicCAS(addr, cur, nv);
                if (prev == cur) break;
                cur = prev;
            }
            break;
        }
        case CSTRIPE_AGG_SUM_F64:
            atomicAdd((double *)&dst->f, src.f);
            break;
        default: {
            unsigned long long old =
                atomicAdd((unsigned long long *)&dst->lo, (unsigned long long)src.lo);
            int64_t carry = (old + (unsigned long long)src.lo) < old ? 1 : 0;
            int64_t hi = src.hi + carry;
            if (hi) atomicAdd((unsigned long long *)&dst->hi, (unsigned long long)hi);
            break;
        }
    }
    atomicAdd((unsigned long long *)&dst->cnt, (unsigned long long)src.cnt);
}

__global__ __launch_bounds__(AGG_BLOCK) void grouped_final_kernel(
    const uint32_t *__restrict__ keys_in, const AccCell *__restrict__ cells_in,
    uint32_t n_blocks, uint32_t per_block, uint32_t *__restrict__ keys_out,
    AccCell *__restrict__ cells_out, uint32_t *__restrict__ n_groups_out,
    int *__restrict__ err, const GroupParams gp)
{
    const AggParams &params = gp.base;
    __shared__ int skeys[CSTRIPE_MAX_GROUPS];
    __shared__ ThreadAcc scells[CSTRIPE_MAX_GROUPS * MAX_AGGS];
    for (uint32_t i = threadIdx.x; i < CSTRIPE_MAX_GROUPS; i += AGG_BLOCK) {
        skeys[i] = -1;
        for (uint32_t a = 0; a < params.n_aggs; a++)
            acc_init(scells[i * MAX_AGGS + a], params.aggs[a].kind);
    }
    __syncthreads();

    const uint32_t total = n_blocks * per_block;
    for (uint32_t f = threadIdx.x; f < total; f += AGG_BLOCK) {
        uint32_t k = keys_in[f];
        if (k == 0xFFFFFFFFu) continue;
        int slot = -1;
        for (uint32_t s = 0; s < CSTRIPE_MAX_GROUPS; s++) {
            int old = atomicCAS(&skeys[s], -1, (int)k);
            if (old == -1 || old == (int)k) { slot = (int)s; break; }
        }
        if (slot < 0) { atomicOr(err, 16); continue; }
        const AccCell *src = cells_in + (size_t)f * params.n_aggs;
        for (uint32_t a = 0; a < params.n_aggs; a++)
            cell_merge_atomic(&scells[slot * MAX_AGGS + a], params.aggs[a].kind, src[a]);
    }
    __syncthreads();

    if (threadIdx.x == 0) {
        uint32_t n = 0;
        for (uint32_t s = 0; s < CSTRIPE_MAX_GROUPS; s++) {
            if (skeys[s] < 0) continue;
            keys_out[n] = (uint32_t)skeys[s];
            for (uint32_t a = 0; a < params.n_aggs; a++) {
                const ThreadAcc &t = scells[s * MAX_AGGS + a];
                AccCell c{t.lo, t.hi, t.f, t.cnt};
                cells_out[(size_t)n * params.n_aggs + a] = c;
            }
            n++;
        }
        *n_groups_out = n;
    }
}

/* device-wide reduce over block partials; each launched block reduces a
 * grid-strided slice into out[blockIdx * n_aggs + a]. Large grids (1B-row
 * scans emit ~300k block partials) run this twice: wide pass then a
 * single-block pass over the wide pass's outputs. */
__global__ __launch_bounds__(AGG_BLOCK) void final_reduce_kernel(
    const AccCell *__restrict__ block_in, uint32_t n_blocks,
    AccCell *__restrict__ out, const AggParams params)
{
    __shared__ ThreadAcc lds[AGG_BLOCK / WAVE][MAX_AGGS];
    const uint32_t wid = threadIdx.x / WAVE;
    const uint32_t lane = threadIdx.x % WAVE;
    for (uint32_t a = 0; a < params.n_aggs; a++) {
        ThreadAcc r;
        acc_init(r, params.aggs[a].kind);
        for (uint64_t b = (uint64_t)blockIdx.x * AGG_BLOCK + threadIdx.x;
             b < n_blocks; b += (uint64_t)gridDim.x * AGG_BLOCK) {
            const AccCell c = block_in[b * params.n_aggs + a];
            ThreadAcc t{c.lo, c.hi, c.f, c.cnt};
            acc_merge(r, t, params.aggs[a].kind);
        }
        wave_reduce(r, params.aggs[a].kind);
        if (lane == 0) lds[wid][a] = r;
        __syncthreads();
        if (threadIdx.x == 0) {
            ThreadAcc f = lds[0][a];
            for (uint32_t w = 1; w < AGG_BLOCK / WAVE; w++)
                acc_merge(f, lds[w][a], params.aggs[a].kind);
            AccCell c{f.lo, f.hi, f.f, f.cnt};
            out[(uint64_t)blockIdx.x * params.n_aggs + a] = c;
        }
        __syncthreads();
    }
}

#define FINAL_WIDE_GRID 240

/* launch the (possibly two-stage) final reduce: d_block[n_blocks] -> d_final */
static int launch_final_reduce(cs_gpu_state *g, uint32_t n_blocks,
                               const AggParams &p)
{
    if (n_blocks > 8192) {
        hipLaunchKernelGGL(final_reduce_kernel, dim3(FINAL_WIDE_GRID),
                           dim3(AGG_BLOCK), 0, g->stream,
                           g->d_block, n_blocks, g->d_final2, p);
        HIP_TRY(hipGetLastError());
        hipLaunchKernelGGL(final_reduce_kernel, dim3(1), dim3(AGG_BLOCK), 0,
                           g->stream, g->d_final2, FINAL_WIDE_GRID, g->d_final, p);
        HIP_TRY(hipGetLastError());
    } else {
        hipLaunchKernelGGL(final_reduce_kernel, dim3(1), dim3(AGG_BLOCK), 0,
                           g->stream, g->d_block, n_blocks, g->d_final, p);
        HIP_TRY(hipGetLastError());
    }
    return CSTRIPE_OK;
}

/* closed-form decode of ONE canonical segment to a dense value buffer —
 * used by the batch/parity path (next_batch) for canonical chunks, which
 * the agg kernels never materialize. One thread per value; coalesced
 * stores; the "decode" is just the canonical position formula (format.h). */
__global__ void canon_decode_kernel(const uint8_t *__restrict__ src,
                                    uint8_t *__restrict__ dst,
                                    uint32_t n_values, uint32_t mode,
                                    uint32_t L, uint32_t width, uint64_t hval)
{
    const uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n_values) return;
    if (mode == CSF_SEGMODE_LIT) {
        const uint8_t *s = src + hval + (uint64_t)i * width;
        uint8_t *d = dst + (uint64_t)i * width;
        for (uint32_t j = 0; j < width; j++) d[j] = s[j];
        return;
    }
    if ((mode & 0xF8u) == CSF_SEGMODE_ZR4B_BASE) {      /* width-4 slots */
        uint32_t slot = (uint32_t)hval;
        if (mode != CSF_SEGMODE_ZR4_CONST) {
            const uint32_t pos = 15u + (i == 0 ? L : i + 3u + L);
            slot |= (uint32_t)src[pos] << (8u * L);
        }
        *(uint32_t *)(dst + (uint64_t)i * 4) = slot;
        return;
    }
    uint64_t v;
    if (mode == CSF_SEGMODE_CONST || mode == CSF_SEGMODE_ZR_CONST) {
        v = hval;
    } else {
        const uint32_t pos = canon_pos((uint8_t)mode, L, i);
        uint64_t lo;
        __builtin_memcpy(&lo, src + pos, 8);
        const uint64_t m = (~0ull) >> ((8u - L) * 8u);
        v = (lo & m) | hval;
    }
    *(uint64_t *)(dst + (uint64_t)i * 8) = v;
}

/* restricted-zstd decode: ONE LANE per frame segment. The frames are this
 * writer's own emission (raw literals + predefined-FSE sequences,
 * zstd_r.h), so the decoder needs no per-frame table build — the three
 * spec-fixed FSE tables are staged once. Output goes straight to scratch;
 * the agg kernels then read it like any decoded stream, so zstd timed
 * regions INCLUDE decompression on device (round-1 VERDICT #4). */
__global__ void zr_decode_kernel(const uint8_t *__restrict__ data,
                                 uint8_t *__restrict__ scratch,
                                 const SegDesc *__restrict__ segs, uint32_t n,
                                 const zr_dtables *__restrict__ dt,
                                 int *__restrict__ err)
{
    const uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const SegDesc s = segs[i];
    const int r = zr_decode_frame(data + s.src_off, (int)s.comp_len,
                                  scratch + s.dst_off, (int)s.decomp_len,
                                  dt->ll, dt->ml, dt->of);
    if (r != (int)s.decomp_len) atomicOr(err, 32);
}

/* ---- device-tuned restricted-frame decoder -----------------------------
 * Same algorithm as zstd_r.h zr_decode_frame (which stays the host decoder
 * and the spec reference — parity is pinned lane-for-lane by the GPU
 * tests), restated for a 4-BYTE-ALIGNED LDS-resident frame. The reason it
 * exists: ds_read/ds_write require natural alignment, so every unaligned
 * 8-byte memcpy in the shared decoder compiles to EIGHT ds_read_u8 ops on
 * the serial FSE chain. Here every LDS access is an aligned dword:
 *  - bit-window refill = two ds_read_b32 (frame base is 4-aligned and the
 *    LDS stride leaves >= 8 B slack, so the aligned window may run past
 *    comp_len into the lane's own slack — those junk bits sit above the
 *    marker bit and are never returned);
 *  - FSE cell fetch = one ds_read_b32 (cells are 4 B);
 *  - literal reads = funnel over aligned dwords.
 * Global-memory accesses (decoded output, match copies) keep unaligned
 * 8-byte memcpys — gfx950 global/flat loads are alignment-tolerant and
 * already compile to dwordx2. */
struct zr_brd {
    const uint8_t *src;    /* 4-aligned LDS frame base */
    int32_t bitpos;
    int32_t wbase;         /* 4-aligned byte index of window LSB (-1 empty) */
    uint64_t w;
};

__device__ inline uint32_t zr_lds_u32(const uint8_t *p)   /* p 4-aligned */
{
    return *(const uint32_t *)__builtin_assume_aligned(p, 4);
}

__device__ inline int zr_brd_init(zr_brd *b, const uint8_t *src, int len)
{
    b->src = src;
    b->wbase = -1;
    b->w = 0;
    int last = len - 1;
    while (last >= 0 && src[last] == 0) last--;
    if (last < 0) return -1;
    int hb = 7;
    while (!(src[last] & (1 << hb))) hb--;
    b->bitpos = last * 8 + hb;
    return 0;
}

__device__ inline uint32_t zr_brd_read(zr_brd *b, int nbits)
{
    if (nbits == 0) return 0;
    b->bitpos -= nbits;
    const int32_t bp = b->bitpos < 0 ? 0 : b->bitpos;
    const int32_t byte = bp >> 3;
    if (b->wbase < 0 || byte < b->wbase || (bp - b->wbase * 8) + nbits > 64) {
        int32_t base = (byte - 2) & ~3;     /* byte-base in [2,5] -> any
                                             * <=16-bit read fits the window */
        if (base < 0) base = 0;
        b->w = (uint64_t)zr_lds_u32(b->src + base) |
               ((uint64_t)zr_lds_u32(b->src + base + 4) << 32);
        b->wbase = base;
    }
    const int32_t sh = bp - b->wbase * 8;
    return (uint32_t)(b->w >> sh) & ((1u << nbits) - 1u);
}

/* one 4-byte FSE cell as a single dword: sym | nbits<<8 | base<<16 */
__device__ inline uint32_t zr_cell_d(const zr_dcell *t, uint32_t s)
{
    return *(const uint32_t *)__builtin_assume_aligned(t + s, 4);
}

/* copy n bytes from the LDS frame (arbitrary offset so, frame base
 * 4-aligned with slack) to global dst: aligned-dword funnel loads,
 * unaligned 8-byte global stores */
__device__ inline void zr_lds_copy(uint8_t *dst, const uint8_t *src4,
                                   uint32_t so, uint32_t n)
{
    const uint32_t mis = so & 3u;
    const uint8_t *al = src4 + (so & ~3u);
    uint32_t i = 0;
    if (n >= 8) {
        const uint32_t sh = mis * 8u;
        uint32_t w0 = zr_lds_u32(al);
        for (; i + 8 <= n; i += 8) {
            uint32_t w1 = zr_lds_u32(al + i + 4);
            uint32_t w2 = zr_lds_u32(al + i + 8);
            uint64_t lo = (uint64_t)w0 | ((uint64_t)w1 << 32);
            uint64_t v = sh ? (lo >> sh) | ((uint64_t)w2 << (64 - sh)) : lo;
            __builtin_memcpy(dst + i, &v, 8);
            w0 = w2;
        }
    }
    for (; i < n; i++) dst[i] = src4[so + i];
}

__device__ inline int zr_decode_frame_a4(const uint8_t *src, int slen,
                                         uint8_t *dst, int dcap,
                                         const zr_dcell *llt,
                                         const zr_dcell *mlt,
                                         const zr_dcell *oft)
{
    int ip = 0;
    if (slen < 7) return -1;
    if (!(src[0] == 0x28 && src[1] == 0xB5 && src[2] == 0x2F && src[3] == 0xFD))
        return -1;
    ip = 4;
    const uint8_t fhd = src[ip++];
    if (!(fhd == 0x20 || fhd == 0x60 || fhd == 0xA0)) return -1;
    int content;
    if (fhd == 0x20) {
        content = src[ip++];
    } else if (fhd == 0x60) {
        content = 256 + src[ip] + (src[ip + 1] << 8);
        ip += 2;
    } else {
        uint32_t f = 0;
        for (int i = 0; i < 4; i++) f |= (uint32_t)src[ip + i] << (8 * i);
        if (f > (1u << 23)) return -1;
        content = (int)f;
        ip += 4;
    }
    if (content > dcap) return -1;
    const uint32_t bh = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8) |
                        ((uint32_t)src[ip + 2] << 16);
    ip += 3;
    const int btype = (int)((bh >> 1) & 3);
    const int bsize = (int)(bh >> 3);
    if (ip + bsize > slen) return -1;
    if (btype == 0) {                     /* raw block */
        if (bsize != content) return -1;
        zr_lds_copy(dst, src, (uint32_t)ip, (uint32_t)content);
        return content;
    }
    if (btype != 2) return -1;
    const int bend = ip + bsize;

    const uint8_t lh = src[ip];           /* literals: RAW only */
    if ((lh & 3) != 0) return -1;
    int lit_size;
    if (!(lh & 0x04)) { lit_size = lh >> 3; ip += 1; }
    else if (!(lh & 0x08)) { lit_size = (lh >> 4) | ((int)src[ip + 1] << 4); ip += 2; }
    else { lit_size = (lh >> 4) | ((int)src[ip + 1] << 4) | ((int)src[ip + 2] << 12); ip += 3; }
    const uint32_t lit_off = (uint32_t)ip;
    ip += lit_size;
    if (ip > bend) return -1;

    int nseq = src[ip++];
    if (nseq >= 128) {
        if (nseq == 255) { nseq = src[ip] + (src[ip + 1] << 8) + 0x7F00; ip += 2; }
        else { nseq = ((nseq - 128) << 8) + src[ip]; ip += 1; }
    }
    int op = 0, lp = 0;
    if (nseq == 0) {
        if (lit_size != content) return -1;
        zr_lds_copy(dst, src, lit_off, (uint32_t)content);
        return content;
    }
    if (src[ip++] != 0x00) return -1;

    zr_brd br;
    if (zr_brd_init(&br, src + ip, bend - ip) < 0) return -1;
    uint32_t sll = zr_brd_read(&br, ZR_LL_ACCLOG);
    uint32_t sof = zr_brd_read(&br, ZR_OF_ACCLOG);
    uint32_t sml = zr_brd_read(&br, ZR_ML_ACCLOG);

    for (int n = 0; n < nseq; n++) {
        const uint32_t cll = zr_cell_d(llt, sll);
        const uint32_t cml = zr_cell_d(mlt, sml);
        const uint32_t cof = zr_cell_d(oft, sof);
        const uint32_t of_sym = cof & 0xFF, ml_sym = cml & 0xFF, ll_sym = cll & 0xFF;
        const uint32_t of_val = (1u << of_sym) + zr_brd_read(&br, (int)of_sym);
        uint32_t ml = ZR_ML_BASE[ml_sym] + zr_brd_read(&br, ZR_ML_BITS[ml_sym]);
        uint32_t ll = ZR_LL_BASE[ll_sym] + zr_brd_read(&br, ZR_LL_BITS[ll_sym]);
        if (n + 1 < nseq) {
            sll = (cll >> 16) + zr_brd_read(&br, (int)((cll >> 8) & 0xFF));
            sml = (cml >> 16) + zr_brd_read(&br, (int)((cml >> 8) & 0xFF));
            sof = (cof >> 16) + zr_brd_read(&br, (int)((cof >> 8) & 0xFF));
        }
        if (of_val <= 3) return -1;
        const uint32_t off = of_val - 3;
        if (lp + (int)ll > lit_size || op + (int)(ll + ml) > content) return -1;
        zr_lds_copy(dst + op, src, lit_off + (uint32_t)lp, ll);
        op += (int)ll; lp += (int)ll;
        if (off > (uint32_t)op) return -1;
        {   /* match: global->global, unaligned 8B memcpys are fine there */
            uint32_t i = 0;
            if (off >= 8) {
                for (; i + 8 <= ml; i += 8) {
                    uint64_t v;
                    __builtin_memcpy(&v, dst + op + (int)i - (int)off, 8);
                    __builtin_memcpy(dst + op + (int)i, &v, 8);
                }
            }
            for (; i < ml; i++) dst[op + (int)i] = dst[op + (int)i - (int)off];
            op += (int)ml;
        }
    }
    const int rem = lit_size - lp;
    if (op + rem != content) return -1;
    zr_lds_copy(dst + op, src, lit_off + (uint32_t)lp, (uint32_t)rem);
    return content;
}

/* LDS-staged variant: the global kernel's wall is the serial FSE decode's
 * dependent byte loads — each lane walks its own frame, the block's frame
 * working set (~128 x 512 B) blows past L1, so every window refill / literal
 * read is an uncoalesced L2-or-HBM round trip. Here the block first copies
 * its 128 frames into LDS with coalesced dword loads (and the 640 B spec
 * FSE tables once), then each lane decodes ITS frame entirely out of LDS
 * via the aligned-dword decoder above; only the decoded output goes to
 * global. Chosen by launch_decode when the largest frame fits the LDS
 * budget (writer default 512 B zstd segments -> ~72 KB/block, 2 blocks/CU). */
#define ZR_LDS_BLOCK 128
__global__ __launch_bounds__(ZR_LDS_BLOCK) void zr_decode_lds_kernel(
    const uint8_t *__restrict__ data, uint8_t *__restrict__ scratch,
    const SegDesc *__restrict__ segs, uint32_t n, uint32_t stride,
    const zr_dtables *__restrict__ dt, int *__restrict__ err)
{
    extern __shared__ uint8_t zbuf[];   /* [ZR_LDS_BLOCK*stride] | tables */
    zr_dcell *tll = (zr_dcell *)(zbuf + (size_t)ZR_LDS_BLOCK * stride);
    zr_dcell *tml = tll + (1 << ZR_LL_ACCLOG);
    zr_dcell *tof = tml + (1 << ZR_ML_ACCLOG);
    __shared__ uint64_t s_src[ZR_LDS_BLOCK];
    __shared__ uint32_t s_len[ZR_LDS_BLOCK];
    {   /* spec tables: 640 B of 4 B cells, cooperative copy */
        const uint32_t *ts = (const uint32_t *)dt;
        uint32_t *td = (uint32_t *)tll;
        for (uint32_t j = threadIdx.x; j < sizeof(zr_dtables) / 4; j += ZR_LDS_BLOCK)
            td[j] = ts[j];
    }
    const uint32_t base_i = blockIdx.x * ZR_LDS_BLOCK;
    const uint32_t i = base_i + threadIdx.x;
    if (i < n) { s_src[threadIdx.x] = segs[i].src_off; s_len[threadIdx.x] = segs[i].comp_len; }
    else s_len[threadIdx.x] = 0;
    __syncthreads();
    const uint32_t nf = n - base_i < ZR_LDS_BLOCK ? n - base_i : ZR_LDS_BLOCK;
    for (uint32_t f = 0; f < nf; f++) {
        /* coalesced staging: unaligned global dword loads (hardware-
         * tolerated), aligned LDS dword writes */
        const uint8_t *src = data + s_src[f];
        uint32_t *d4 = (uint32_t *)(zbuf + (size_t)f * stride);
        const uint32_t nw = (s_len[f] + 3) >> 2;
        for (uint32_t j = threadIdx.x; j < nw; j += ZR_LDS_BLOCK) {
            uint32_t v;
            __builtin_memcpy(&v, src + 4 * j, 4);
            d4[j] = v;
        }
    }
    __syncthreads();
    if (i >= n) return;
    const SegDesc s = segs[i];
    const int r = zr_decode_frame_a4(zbuf + (size_t)threadIdx.x * stride, (int)s.comp_len,
                                     scratch + s.dst_off, (int)s.decomp_len,
                                     tll, tml, tof);
    if (r != (int)s.decomp_len) atomicOr(err, 32);
}

/* LDS-in + LDS-out variant: PMC on the kernel above shows 86% of wave
 * cycles parked on s_waitcnt — with the frames LDS-resident, the residual
 * latency chain is the OUTPUT path: every match copy is a load-after-store
 * round trip through global L2 (stores bypass L1 on CDNA), serialized by
 * vmcnt waits 8 bytes at a time. Here each lane decodes into its own LDS
 * output slot, so match copies stay in LDS, and the block flushes decoded
 * slots to global with coalesced 16 B stores at the end. Costs ~2x the LDS
 * (1 block/CU at the 512 B writer default) but removes the global round
 * trips from the serial chain. */
__global__ __launch_bounds__(ZR_LDS_BLOCK) void zr_decode_lds2_kernel(
    const uint8_t *__restrict__ data, uint8_t *__restrict__ scratch,
    const SegDesc *__restrict__ segs, uint32_t n, uint32_t stride,
    uint32_t ostride, const zr_dtables *__restrict__ dt, int *__restrict__ err)
{
    extern __shared__ uint8_t zbuf[];   /* [in frames][out slots][tables] */
    uint8_t *obuf = zbuf + (size_t)ZR_LDS_BLOCK * stride;
    zr_dcell *tll = (zr_dcell *)(obuf + (size_t)ZR_LDS_BLOCK * ostride);
    zr_dcell *tml = tll + (1 << ZR_LL_ACCLOG);
    zr_dcell *tof = tml + (1 << ZR_ML_ACCLOG);
    __shared__ uint64_t s_src[ZR_LDS_BLOCK];
    __shared__ uint64_t s_dst[ZR_LDS_BLOCK];
    __shared__ uint32_t s_len[ZR_LDS_BLOCK];
    __shared__ uint32_t s_dlen[ZR_LDS_BLOCK];
    {
        const uint32_t *ts = (const uint32_t *)dt;
        uint32_t *td = (uint32_t *)tll;
        for (uint32_t j = threadIdx.x; j < sizeof(zr_dtables) / 4; j += ZR_LDS_BLOCK)
            td[j] = ts[j];
    }
    const uint32_t base_i = blockIdx.x * ZR_LDS_BLOCK;
    const uint32_t i = base_i + threadIdx.x;
    if (i < n) {
        const SegDesc s = segs[i];
        s_src[threadIdx.x] = s.src_off;
        s_dst[threadIdx.x] = s.dst_off;
        s_len[threadIdx.x] = s.comp_len;
        s_dlen[threadIdx.x] = s.decomp_len;
    } else { s_len[threadIdx.x] = 0; s_dlen[threadIdx.x] = 0; }
    __syncthreads();
    const uint32_t nf = n - base_i < ZR_LDS_BLOCK ? n - base_i : ZR_LDS_BLOCK;
    for (uint32_t f = 0; f < nf; f++) {
        const uint8_t *src = data + s_src[f];
        uint32_t *d4 = (uint32_t *)(zbuf + (size_t)f * stride);
        const uint32_t nw = (s_len[f] + 3) >> 2;
        for (uint32_t j = threadIdx.x; j < nw; j += ZR_LDS_BLOCK) {
            uint32_t v;
            __builtin_memcpy(&v, src + 4 * j, 4);
            d4[j] = v;
        }
    }
    __syncthreads();
    if (i < n) {
        const int r = zr_decode_frame_a4(zbuf + (size_t)threadIdx.x * stride,
                                         (int)s_len[threadIdx.x],
                                         obuf + (size_t)threadIdx.x * ostride,
                                         (int)s_dlen[threadIdx.x],
                                         tll, tml, tof);
        if (r != (int)s_dlen[threadIdx.x]) atomicOr(err, 32);
    }
    __syncthreads();
    for (uint32_t f = 0; f < nf; f++) {     /* coalesced flush, exact dlen */
        uint8_t *dst = scratch + s_dst[f];
        const uint8_t *out = obuf + (size_t)f * ostride;   /* 4-aligned slot */
        const uint32_t dlen = s_dlen[f];
        const uint32_t vec = dlen >> 2;
        for (uint32_t j = threadIdx.x; j < vec; j += ZR_LDS_BLOCK) {
            const uint32_t v = zr_lds_u32(out + 4 * j);
            __builtin_memcpy(dst + 4 * j, &v, 4);   /* global: unaligned ok */
        }
        for (uint32_t j = (vec << 2) + threadIdx.x; j < dlen; j += ZR_LDS_BLOCK)
            dst[j] = out[j];
    }
}

/* =====================================================================
 * device chunk-group pruning (SelectedChunkMask on GPU — SURVEY §8f3)
 * One thread per chunk group evaluates the same CNF refutation as the
 * host loop in cstripe_scan_begin (reference: SelectedChunkMask,
 * columnar_reader.c:1132-1187; OR recursion columnar_customscan.c:770-829),
 * including PG float ordering and the all-NULL-chunk rule. The host only
 * GATHERS the skip entries (a copy pass, no comparisons); every compare
 * runs on device.
 * ===================================================================== */

struct PruneMM { int64_t mn, mx; uint64_t has; };

struct PrunePred {
    int64_t ival;
    double fval;
    uint8_t op, is_float, slot, gend;
};

struct PruneParams {
    uint32_t n_preds;
    PrunePred preds[CSTRIPE_MAX_PREDS];
};

__global__ void chunk_prune_kernel(const PruneMM *__restrict__ mm,
                                   uint32_t n_chunks,
                                   uint8_t *__restrict__ selected,
                                   const PruneParams pp)
{
    const uint32_t c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= n_chunks) return;
    bool sel = true;
    bool group_refuted = true;
    for (uint32_t i = 0; i < pp.n_preds; i++) {
        const PrunePred &p = pp.preds[i];
        const PruneMM e = mm[(size_t)p.slot * n_chunks + c];
        bool ref = false;
        if (e.has) {          /* all-NULL chunks are never refuted */
            if (p.is_float) {
                double mn, mx;
                __builtin_memcpy(&mn, &e.mn, 8);
                __builtin_memcpy(&mx, &e.mx, 8);
                const double cv = p.fval;
                switch (p.op) {     /* PG float ordering (NaN high) */
                    case CSTRIPE_PRED_LT: ref = f64cmp_pg(mn, cv) >= 0; break;
                    case CSTRIPE_PRED_LE: ref = f64cmp_pg(mn, cv) > 0; break;
                    case CSTRIPE_PRED_GT: ref = f64cmp_pg(mx, cv) <= 0; break;
                    case CSTRIPE_PRED_GE: ref = f64cmp_pg(mx, cv) < 0; break;
                    case CSTRIPE_PRED_EQ: ref = f64cmp_pg(cv, mn) < 0 ||
                                                f64cmp_pg(cv, mx) > 0; break;
                    default:              ref = f64cmp_pg(mn, cv) == 0 &&
                                                f64cmp_pg(mx, cv) == 0; break;
                }
            } else {
                const int64_t cv = p.ival;
                switch (p.op) {
                    case CSTRIPE_PRED_LT: ref = e.mn >= cv; break;
                    case CSTRIPE_PRED_LE: ref = e.mn > cv; break;
                    case CSTRIPE_PRED_GT: ref = e.mx <= cv; break;
                    case CSTRIPE_PRED_GE: ref = e.mx < cv; break;
                    case CSTRIPE_PRED_EQ: ref = cv < e.mn || cv > e.mx; break;
                    default:              ref = e.mn == cv && e.mx == cv; break;
                }
            }
        }
        group_refuted &= ref;
        if (p.gend) {
            if (group_refuted) { sel = false; break; }
            group_refuted = true;
        }
    }
    selected[c] = sel ? 1 : 0;
}

int csgpu_prune(cstripe_reader *r, const std::vector<cstripe_pred> &preds,
                std::vector<uint8_t> &selected)
{
    if (!cstripe_gpu_available()) return CSTRIPE_ERR_NOGPU;
    if (preds.empty() || preds.size() > CSTRIPE_MAX_PREDS) return CSTRIPE_ERR_ARG;

    /* distinct predicate columns -> slots */
    PruneParams pp{};
    pp.n_preds = (uint32_t)preds.size();
    uint32_t slot_col[CSTRIPE_MAX_PREDS];
    uint32_t n_slots = 0;
    for (uint32_t i = 0; i < pp.n_preds; i++) {
        const cstripe_pred &p = preds[i];
        uint32_t sl = n_slots;
        for (uint32_t j = 0; j < n_slots; j++)
            if (slot_col[j] == p.column) { sl = j; break; }
        if (sl == n_slots) slot_col[n_slots++] = p.column;
        const uint8_t ty = r->cols[p.column].type;
        pp.preds[i].ival = p.ival;
        pp.preds[i].fval = p.fval;
        pp.preds[i].op = p.op;
        pp.preds[i].is_float = ty == CSTRIPE_F32 || ty == CSTRIPE_F64;
        pp.preds[i].slot = (uint8_t)sl;
        pp.preds[i].gend = i + 1 == pp.n_preds ||
                           preds[i + 1].or_group != preds[i].or_group;
    }

    uint64_t n_chunks = 0;
    for (const auto &st : r->stripes) n_chunks += st.meta.chunk_count;
    if (n_chunks == 0) { selected.clear(); return CSTRIPE_OK; }

    /* gather skip entries per slot, global chunk order */
    std::vector<PruneMM> h_mm((size_t)n_slots * n_chunks);
    uint64_t ci = 0;
    for (const auto &st : r->stripes) {
        for (uint32_t k = 0; k < st.meta.chunk_count; k++, ci++) {
            for (uint32_t sl = 0; sl < n_slots; sl++) {
                const csf_skipnode &nd = st.nodes[slot_col[sl]][k].n;
                PruneMM &e = h_mm[(size_t)sl * n_chunks + ci];
                e.mn = nd.min_i;
                e.mx = nd.max_i;
                e.has = nd.has_min_max;
            }
        }
    }

    PruneMM *d_mm = nullptr;
    uint8_t *d_sel = nullptr;
    #define PR_TRY(x) do { hipError_t _e = (x); if (_e != hipSuccess) { \
        cs_set_err("prune: %s", hipGetErrorString(_e)); \
        if (d_mm) (void)hipFree(d_mm); if (d_sel) (void)hipFree(d_sel); \
        return CSTRIPE_ERR; } } while (0)
    PR_TRY(hipMalloc(&d_mm, h_mm.size() * sizeof(PruneMM)));
    PR_TRY(hipMalloc(&d_sel, n_chunks));
    PR_TRY(hipMemcpy(d_mm, h_mm.data(), h_mm.size() * sizeof(PruneMM),
                     hipMemcpyHostToDevice));
    const uint32_t grid = (uint32_t)((n_chunks + 255) / 256);
    hipLaunchKernelGGL(chunk_prune_kernel, dim3(grid), dim3(256), 0, 0,
                       d_mm, (uint32_t)n_chunks, d_sel, pp);
    PR_TRY(hipGetLastError());
    selected.resize(n_chunks);
    PR_TRY(hipMemcpy(selected.data(), d_sel, n_chunks, hipMemcpyDeviceToHost));
    #undef PR_TRY
    (void)hipFree(d_mm);
    (void)hipFree(d_sel);
    return CSTRIPE_OK;
}

/* =====================================================================
 * staging
 * ===================================================================== */

static uint64_t align_up(uint64_t x, uint64_t a) { return (x + a - 1) & ~(a - 1); }

/* single-segment chunk with a CLOSED-FORM canonical stream (lz4 P/CONST/LIT
 * or restricted-zstd P/CONST, format.h): staged compressed, read in place */
static inline bool cs_node_canon(const cs_skipnode &nd)
{
    if (nd.n.n_segs != 1) return false;
    const uint8_t m = nd.seg_modes[0];
    if (nd.n.comp_type == CSTRIPE_COMP_LZ4)
        return m != CSF_SEGMODE_GENERIC;
    if (nd.n.comp_type == CSTRIPE_COMP_ZSTD)
        return m == CSF_SEGMODE_ZR_CONST ||
               (m > CSF_SEGMODE_ZRP_BASE && m <= CSF_SEGMODE_ZRP_BASE + 4) ||
               (m >= CSF_SEGMODE_ZR4B_BASE && m <= CSF_SEGMODE_ZR4_CONST);
    return false;
}

void csgpu_release(cstripe_scan *s)
{
    if (!s || !s->gpu) return;
    cs_gpu_state *g = s->gpu;
    #define HIP_DROP(x) do { hipError_t _e = (x); (void)_e; } while (0)
    if (g->d_data) HIP_DROP(hipFree(g->d_data));
    if (g->d_scratch) HIP_DROP(hipFree(g->d_scratch));
    if (g->d_rank) HIP_DROP(hipFree(g->d_rank));
    if (g->d_segs) HIP_DROP(hipFree(g->d_segs));
    if (g->d_groups) HIP_DROP(hipFree(g->d_groups));
    if (g->d_colloc) HIP_DROP(hipFree(g->d_colloc));
    if (g->d_block) HIP_DROP(hipFree(g->d_block));
    if (g->d_final) HIP_DROP(hipFree(g->d_final));
    if (g->d_final2) HIP_DROP(hipFree(g->d_final2));
    if (g->d_gkeys) HIP_DROP(hipFree(g->d_gkeys));
    if (g->d_gcells) HIP_DROP(hipFree(g->d_gcells));
    if (g->d_gfkeys) HIP_DROP(hipFree(g->d_gfkeys));
    if (g->d_gfcells) HIP_DROP(hipFree(g->d_gfcells));
    if (g->d_gn) HIP_DROP(hipFree(g->d_gn));
    if (g->d_tmp) HIP_DROP(hipFree(g->d_tmp));
    if (g->d_zsegs) HIP_DROP(hipFree(g->d_zsegs));
    if (g->d_zrtab) HIP_DROP(hipFree(g->d_zrtab));
    if (g->d_segstart) HIP_DROP(hipFree(g->d_segstart));
    if (g->d_tiles) HIP_DROP(hipFree(g->d_tiles));
    if (g->d_tiles2) HIP_DROP(hipFree(g->d_tiles2));
    if (g->d_error) HIP_DROP(hipFree(g->d_error));
    if (g->ev0) HIP_DROP(hipEventDestroy(g->ev0));
    if (g->ev1) HIP_DROP(hipEventDestroy(g->ev1));
    if (g->ev2) HIP_DROP(hipEventDestroy(g->ev2));
    if (g->stream) HIP_DROP(hipStreamDestroy(g->stream));
    #undef HIP_DROP
    delete g;
    s->gpu = nullptr;
}

uint64_t csgpu_staged_bytes(const cstripe_scan *s)
{
    return s->gpu ? s->gpu->data_bytes : 0;
}

int csgpu_stage(cstripe_scan *s, int device_id)
{
    if (!cstripe_gpu_available()) { cs_set_err("no HIP device visible — the cstripe GPU path requires an MI355X (no CPU fallback)"); return CSTRIPE_ERR_NOGPU; }
    if (s->gpu) csgpu_release(s);

    cstripe_reader *r = s->r;
    auto *g = new cs_gpu_state();
    s->gpu = g;
    if (device_id >= 0) {
        if (hipSetDevice(device_id) != hipSuccess) { cs_set_err("hipSetDevice(%d) failed", device_id); csgpu_release(s); return CSTRIPE_ERR; }
        g->device = device_id;
    } else {
        (void)hipGetDevice(&g->device);
    }

    /* projected column slots */
    for (int i = 0; i < 64; i++) g->proj_of_col[i] = -1;
    uint32_t n_proj = 0;
    for (uint32_t c = 0; c < r->head.column_count; c++) {
        if (s->cols_mask & (1ull << c)) {
            if (n_proj >= MAX_PROJ) { cs_set_err("too many projected columns"); csgpu_release(s); return CSTRIPE_ERR_ARG; }
            g->proj_of_col[c] = (int)n_proj;
            g->proj_type[n_proj] = r->cols[c].type;
            n_proj++;
        }
    }
    if (n_proj == 0) { cs_set_err("empty projection"); csgpu_release(s); return CSTRIPE_ERR_ARG; }
    g->n_proj = n_proj;
    g->n_groups = (uint32_t)s->sel.size();

    /* host pass 1: compute layout sizes */
    uint64_t data_bytes = 0, scratch_bytes = 0, rank_words = 0;
    uint32_t n_segs = 0, n_zsegs = 0;
    bool zstd_host = false;
    for (const auto &sc : s->sel) {
        const cs_stripe_info &st = r->stripes[sc.stripe];
        uint32_t rows = st.group_rows[sc.chunk];
        for (uint32_t c = 0; c < r->head.column_count; c++) {
            if (g->proj_of_col[c] < 0) continue;
            const cs_skipnode &nd = st.nodes[c][sc.chunk];
            data_bytes = align_up(data_bytes, 8) + align_up((rows + 7) / 8, 8); /* exists, 8B padded */
            const bool canon = cs_node_canon(nd);
            bool zr_dev = !canon && nd.n.comp_type == CSTRIPE_COMP_ZSTD;
            if (zr_dev)
                for (uint8_t m : nd.seg_modes)
                    if (m != CSF_SEGMODE_ZR) { zr_dev = false; break; }
            if (canon) {
                /* canonical stream: values are read closed-form from the
                 * compressed bytes — no scratch, no decode segments */
                data_bytes = align_up(data_bytes, 16) + align_up(nd.n.value_len, 16);
            } else if (zr_dev) {
                /* restricted-zstd frames: staged compressed, decoded on
                 * device into scratch (zr_decode_kernel) */
                data_bytes = align_up(data_bytes, 16) + align_up(nd.n.value_len, 16);
                scratch_bytes = align_up(scratch_bytes, 16) + align_up(nd.n.decompressed_size, 16);
                n_zsegs += nd.n.n_segs;
            } else if (nd.n.comp_type == CSTRIPE_COMP_LZ4) {
                data_bytes = align_up(data_bytes, 16) + align_up(nd.n.value_len, 16);
                scratch_bytes = align_up(scratch_bytes, 16) + align_up(nd.n.decompressed_size, 16);
                n_segs += nd.n.n_segs;
            } else { /* NONE staged raw; ZSTD host-decoded at stage (documented fallback) */
                if (nd.n.comp_type == CSTRIPE_COMP_ZSTD || nd.n.comp_type == CSTRIPE_COMP_PGLZ) zstd_host = true;
                data_bytes = align_up(data_bytes, 16) + align_up(nd.n.decompressed_size, 16);
            }
            if (nd.n.n_present != nd.n.row_count)
                rank_words += (rows + 63) / 64;
        }
    }
    (void)zstd_host;

    uint32_t tiles_pg = (r->head.chunk_row_limit + TILE_ROWS - 1) / TILE_ROWS;
    if (tiles_pg == 0) tiles_pg = 1;
    uint64_t n_tiles_max = 0;
    for (const auto &sc : s->sel)
        n_tiles_max += (r->stripes[sc.stripe].group_rows[sc.chunk] + 2047) / 2048;
    uint64_t max_blocks = (uint64_t)g->n_groups * tiles_pg;
    if (n_tiles_max > max_blocks) max_blocks = n_tiles_max;
    {   /* lds_agg_kernel tiles every chunk at the full 2048-row pitch */
        uint64_t b2 = (uint64_t)g->n_groups *
                      ((r->head.chunk_row_limit + 2047) / 2048);
        if (b2 > max_blocks) max_blocks = b2;
    }
    g->max_blocks = (uint32_t)max_blocks;

    HIP_TRY(hipStreamCreate(&g->stream));
    HIP_TRY(hipEventCreate(&g->ev0));
    HIP_TRY(hipEventCreate(&g->ev1));
    HIP_TRY(hipEventCreate(&g->ev2));
    data_bytes += 128;              /* window read slack: R-row extraction
                                     * windows reach up to ~60 B past the
                                     * last region's stream end */
    if (data_bytes) HIP_TRY(hipMalloc(&g->d_data, data_bytes));
    if (scratch_bytes) HIP_TRY(hipMalloc(&g->d_scratch, scratch_bytes));
    if (rank_words) HIP_TRY(hipMalloc(&g->d_rank, rank_words * 4));
    if (n_segs) HIP_TRY(hipMalloc(&g->d_segs, (uint64_t)n_segs * sizeof(SegDesc)));
    if (n_zsegs) {
        HIP_TRY(hipMalloc(&g->d_zsegs, (uint64_t)n_zsegs * sizeof(SegDesc)));
        HIP_TRY(hipMalloc(&g->d_zrtab, sizeof(zr_dtables)));
    }
    HIP_TRY(hipMalloc(&g->d_groups, (uint64_t)g->n_groups * sizeof(GroupDesc)));
    HIP_TRY(hipMalloc(&g->d_colloc, (uint64_t)g->n_groups * n_proj * sizeof(ColLoc)));
    HIP_TRY(hipMalloc(&g->d_block, max_blocks * MAX_AGGS * sizeof(AccCell)));
    HIP_TRY(hipMalloc(&g->d_final, MAX_AGGS * sizeof(AccCell)));
    HIP_TRY(hipMalloc(&g->d_final2, (uint64_t)240 * MAX_AGGS * sizeof(AccCell)));
    HIP_TRY(hipMalloc(&g->d_error, sizeof(int)));
    g->data_bytes = data_bytes;
    g->scratch_bytes = scratch_bytes;
    g->n_segs = n_segs;
    g->n_zsegs = n_zsegs;

    /* host pass 2: build staging buffer + descriptors */
    std::vector<uint8_t> h_data(data_bytes, 0);
    std::vector<uint32_t> h_rank;
    h_rank.reserve(rank_words);
    std::vector<SegDesc> h_segs;
    h_segs.reserve(n_segs);
    std::vector<SegDesc> h_zsegs;
    h_zsegs.reserve(n_zsegs);
    std::vector<uint32_t> seg_start((uint64_t)g->n_groups * n_proj, 0);
    std::vector<uint8_t> col_g256(n_proj, 1);   /* stays 1 if every chunk is
                                                 * dense uniform-256B lz4 */
    g->col_scratch.assign(n_proj, 0);
    if (n_proj > 4) g->fusable = false;
    std::vector<GroupDesc> h_groups(g->n_groups);
    std::vector<ColLoc> h_colloc((uint64_t)g->n_groups * n_proj);
    g->scratch_off.assign((uint64_t)g->n_groups * n_proj, ~0ull);

    uint64_t dpos = 0, spos = 0;
    std::vector<uint8_t> tmp;   /* zstd host-decode scratch */
    for (uint32_t gi = 0; gi < g->n_groups; gi++) {
        const cs_selchunk &sc = s->sel[gi];
        const cs_stripe_info &st = r->stripes[sc.stripe];
        uint32_t rows = st.group_rows[sc.chunk];
        h_groups[gi].row_count = rows;
        h_groups[gi].colbase = gi * n_proj;
        const uint8_t *stripe_base = r->stripe_base(st);
        for (uint32_t c = 0; c < r->head.column_count; c++) {
            int pj = g->proj_of_col[c];
            if (pj < 0) continue;
            const cs_skipnode &nd = st.nodes[c][sc.chunk];
            ColLoc cl{};
            cl.type = r->cols[c].type;
            cl.width = (uint8_t)csf_type_width(cl.type);
            bool dense = (nd.n.n_present == nd.n.row_count);

            /* exists bitmap (8B padded) */
            dpos = align_up(dpos, 8);
            cl.exists_off = dpos;
            memcpy(h_data.data() + dpos, stripe_base + nd.n.exists_off, nd.n.exists_len);
            dpos += align_up((rows + 7) / 8, 8);

            if (!dense) {
                cl.rank_off = (uint32_t)h_rank.size();
                uint32_t words = (rows + 63) / 64;
                const uint8_t *eb = stripe_base + nd.n.exists_off;
                uint32_t run = 0;
                for (uint32_t w = 0; w < words; w++) {
                    h_rank.push_back(run);
                    uint64_t wv = 0;
                    uint32_t nb = (uint32_t)nd.n.exists_len - w * 8 >= 8 ? 8 : (uint32_t)nd.n.exists_len - w * 8;
                    memcpy(&wv, eb + w * 8, nb);
                    run += (uint32_t)__builtin_popcountll(wv);
                }
            } else {
                cl.flags |= 2;
            }
            if (!(cl.flags & 2)) g->all_dense = false;
            {
                const cs_skipnode &nd2 = st.nodes[c][sc.chunk];
                const bool cp = nd2.n.comp_type == CSTRIPE_COMP_LZ4 &&
                                nd2.n.n_segs == 1 &&
                                (nd2.seg_modes[0] == CSF_SEGMODE_CONST ||
                                 (nd2.seg_modes[0] > CSF_SEGMODE_P_BASE &&
                                  nd2.seg_modes[0] <= CSF_SEGMODE_P_BASE + 4));
                if (!cp || !(cl.flags & 2)) g->all_canonP = false;
            }

            const bool canon = cs_node_canon(nd);
            if (canon || !(cl.flags & 2) || cl.type != CSTRIPE_I64 ||
                nd.n.comp_type != CSTRIPE_COMP_LZ4)
                g->fusable = false;
            if (canon || !(cl.flags & 2) || (cl.width != 8 && cl.width != 1) ||
                nd.n.comp_type != CSTRIPE_COMP_LZ4)
                g->fusable_mixed = false;
            bool zr_dev = !canon && nd.n.comp_type == CSTRIPE_COMP_ZSTD;
            if (zr_dev)
                for (uint8_t m : nd.seg_modes)
                    if (m != CSF_SEGMODE_ZR) { zr_dev = false; break; }
            if (zr_dev) {
                col_g256[pj] = 0;
                g->col_scratch[pj] = 1;
                dpos = align_up(dpos, 16);
                memcpy(h_data.data() + dpos, stripe_base + nd.n.value_off, nd.n.value_len);
                spos = align_up(spos, 16);
                cl.val_off = spos;
                cl.flags |= 1;                    /* values land in scratch */
                g->scratch_off[(uint64_t)gi * n_proj + pj] = spos;
                for (const csf_seg &sg : st.nodes[c][sc.chunk].segs) {
                    SegDesc sd;
                    sd.src_off = dpos + sg.comp_off;
                    sd.dst_off = spos + sg.decomp_off;
                    sd.comp_len = sg.comp_len;
                    sd.decomp_len = sg.decomp_len;
                    if (sg.comp_len > g->max_zseg_comp) g->max_zseg_comp = sg.comp_len;
                    if (sd.decomp_len > g->max_zseg_dlen) g->max_zseg_dlen = sd.decomp_len;
                    h_zsegs.push_back(sd);
                }
                dpos += align_up(nd.n.value_len, 16);
                spos += align_up(nd.n.decompressed_size, 16);
            } else if (canon) {
                col_g256[pj] = 0;
                /* stage the compressed stream; kernels read values straight
                 * out of it (col_value canonical path) */
                dpos = align_up(dpos, 16);
                cl.val_off = dpos;
                memcpy(h_data.data() + dpos, stripe_base + nd.n.value_off, nd.n.value_len);
                dpos += align_up(nd.n.value_len, 16);
                cl.flags |= 4;
                cl.mode = nd.seg_modes[0];
                const uint8_t *strm = stripe_base + nd.n.value_off;
                if (cl.mode == CSF_SEGMODE_LIT) {
                    cl.hval = (int64_t)csf_canon_lit_hdr(nd.segs[0].decomp_len);
                } else if ((cl.mode & 0xF8u) == CSF_SEGMODE_ZR4B_BASE) {
                    uint32_t s0;
                    memcpy(&s0, strm + 15, 4);       /* slot0 at lit offset 0 */
                    if (cl.mode == CSF_SEGMODE_ZR4_CONST) {
                        cl.hval = (int64_t)s0;
                    } else {
                        cl.L = (uint8_t)(cl.mode & 0x3);
                        cl.hval = (int64_t)(s0 & ~(0xFFu << (8u * cl.L)));
                    }
                } else {
                    /* v0 full: first literal byte — offset 1 in a canonical
                     * LZ4 block (token first), 15 in a canonical zstd frame
                     * (fixed header, format.h) */
                    const bool zrm = cl.mode >= CSF_SEGMODE_ZRP_BASE;
                    uint64_t v0;
                    memcpy(&v0, strm + (zrm ? 15 : 1), 8);
                    if (cl.mode == CSF_SEGMODE_CONST ||
                        cl.mode == CSF_SEGMODE_ZR_CONST) {
                        cl.hval = (int64_t)v0;
                    } else {                      /* P(L) */
                        cl.L = (uint8_t)(cl.mode & 0xF);
                        const uint64_t m = (~0ull) >> ((8u - cl.L) * 8u);
                        cl.hval = (int64_t)(v0 & ~m);
                    }
                }
            } else if (nd.n.comp_type == CSTRIPE_COMP_LZ4) {
                dpos = align_up(dpos, 16);
                memcpy(h_data.data() + dpos, stripe_base + nd.n.value_off, nd.n.value_len);
                spos = align_up(spos, 16);
                cl.val_off = spos;
                cl.flags |= 1;
                g->scratch_off[(uint64_t)gi * n_proj + pj] = spos;
                seg_start[(uint64_t)gi * n_proj + pj] = (uint32_t)h_segs.size();
                g->col_scratch[pj] = 1;
                if (!(cl.flags & 2)) col_g256[pj] = 0;
                uint32_t segi = 0;
                const auto &seglist = st.nodes[c][sc.chunk].segs;
                for (const csf_seg &sg : seglist) {
                    /* non-final segments exactly 256 B, final <= 256 B: the
                     * fused kernels map row -> lane region as byteoff>>8 and
                     * count tile segments as ceil(bytes/256), so an absorbed
                     * 257-271 B final segment would read the wrong region
                     * and overrun the segment list (round-1 advisor) */
                    if (sg.decomp_off != segi * 256u ||
                        (segi + 1 < seglist.size() ? sg.decomp_len != 256
                                                   : sg.decomp_len > 256)) {
                        g->fusable = false;
                        g->fusable_mixed = false;
                        col_g256[pj] = 0;
                    }
                    segi++;
                }
                for (const csf_seg &sg : st.nodes[c][sc.chunk].segs) {
                    SegDesc sd;
                    sd.src_off = dpos + sg.comp_off;
                    sd.dst_off = spos + sg.decomp_off;
                    sd.comp_len = sg.comp_len;
                    sd.decomp_len = sg.decomp_len;
                    if (sg.comp_len > g->max_seg_comp) g->max_seg_comp = sg.comp_len;
                    if (sg.decomp_len > g->max_seg_dlen) g->max_seg_dlen = sg.decomp_len;
                    if (sg.decomp_off % 16 != 0) g->segs_16aligned = false;
                    h_segs.push_back(sd);
                }
                dpos += align_up(nd.n.value_len, 16);
                spos += align_up(nd.n.decompressed_size, 16);
            } else if (nd.n.comp_type == CSTRIPE_COMP_NONE) {
                col_g256[pj] = 0;
                dpos = align_up(dpos, 16);
                cl.val_off = dpos;
                memcpy(h_data.data() + dpos, stripe_base + nd.n.value_off, nd.n.value_len);
                dpos += align_up(nd.n.value_len, 16);
            } else if (nd.n.comp_type == CSTRIPE_COMP_ZSTD ||
                       nd.n.comp_type == CSTRIPE_COMP_PGLZ) {
                col_g256[pj] = 0;
                /* host-decode at stage (zstd: documented fallback, GPU zstd
                 * is a later item — SURVEY.md §8f(1); pglz is host-only by
                 * design: reference-migrated tables read correctly) */
                dpos = align_up(dpos, 16);
                cl.val_off = dpos;
                tmp.resize(nd.n.decompressed_size);
                for (const csf_seg &sg : st.nodes[c][sc.chunk].segs) {
                    if (nd.n.comp_type == CSTRIPE_COMP_ZSTD) {
                        size_t zr = ZSTD_decompress(tmp.data() + sg.decomp_off, sg.decomp_len,
                                                    stripe_base + nd.n.value_off + sg.comp_off, sg.comp_len);
                        if (ZSTD_isError(zr) || zr != sg.decomp_len) { cs_set_err("zstd host decode failed"); csgpu_release(s); return CSTRIPE_ERR_FORMAT; }
                    } else {
                        const uint8_t *pb = stripe_base + nd.n.value_off + sg.comp_off;
                        if (sg.comp_len < CSPGLZ_HDRSZ ||
                            cspglz_varsize(pb) != sg.comp_len ||
                            cspglz_rawsize(pb) != (int32_t)sg.decomp_len ||
                            cspglz_decompress(pb + CSPGLZ_HDRSZ,
                                              (int32_t)(sg.comp_len - CSPGLZ_HDRSZ),
                                              tmp.data() + sg.decomp_off,
                                              (int32_t)sg.decomp_len) < 0) {
                            cs_set_err("pglz host decode failed");
                            csgpu_release(s);
                            return CSTRIPE_ERR_FORMAT;
                        }
                    }
                }
                memcpy(h_data.data() + dpos, tmp.data(), tmp.size());
                dpos += align_up(nd.n.decompressed_size, 16);
            } else {
                cs_set_err("unsupported chunk compression %d", nd.n.comp_type);
                csgpu_release(s);
                return CSTRIPE_ERR_FORMAT;
            }
            h_colloc[(uint64_t)gi * n_proj + pj] = cl;
        }
    }

    /* invariant: the layout pass and the fill pass must agree exactly —
     * any drift between them is silent corruption (heap overflow on the
     * host copy or device reads past staged regions) */
    if (align_up(dpos, 16) > data_bytes || align_up(spos, 16) > scratch_bytes + 15) {
        cs_set_err("staging layout drift: filled %llu/%llu data, %llu/%llu scratch",
                   (unsigned long long)dpos, (unsigned long long)data_bytes,
                   (unsigned long long)spos, (unsigned long long)scratch_bytes);
        csgpu_release(s);
        return CSTRIPE_ERR;
    }
    HIP_TRY(hipMemcpyAsync(g->d_data, h_data.data(), data_bytes, hipMemcpyHostToDevice, g->stream));
    if (!h_rank.empty())
        HIP_TRY(hipMemcpyAsync(g->d_rank, h_rank.data(), h_rank.size() * 4, hipMemcpyHostToDevice, g->stream));
    if (!h_segs.empty())
        HIP_TRY(hipMemcpyAsync(g->d_segs, h_segs.data(), h_segs.size() * sizeof(SegDesc), hipMemcpyHostToDevice, g->stream));
    if (!h_zsegs.empty()) {
        /* sort frames by compressed length so a wave's 64 lanes decode
         * similar-cost frames: the serial FSE decode runs in lockstep per
         * wave (and per block up to the flush barrier), so one expensive
         * frame otherwise gates 127 cheap ones (frames are independent —
         * dst_off travels with the descriptor) */
        std::stable_sort(h_zsegs.begin(), h_zsegs.end(),
                         [](const SegDesc &a, const SegDesc &b) {
                             return a.comp_len > b.comp_len;
                         });
        HIP_TRY(hipMemcpyAsync(g->d_zsegs, h_zsegs.data(), h_zsegs.size() * sizeof(SegDesc), hipMemcpyHostToDevice, g->stream));
        static zr_dtables h_zrtab;
        static bool zrtab_init = false;
        if (!zrtab_init) { zr_build_dtables(&h_zrtab); zrtab_init = true; }
        HIP_TRY(hipMemcpyAsync(g->d_zrtab, &h_zrtab, sizeof(h_zrtab), hipMemcpyHostToDevice, g->stream));
    }
    HIP_TRY(hipMemcpyAsync(g->d_groups, h_groups.data(), h_groups.size() * sizeof(GroupDesc), hipMemcpyHostToDevice, g->stream));
    HIP_TRY(hipMemcpyAsync(g->d_colloc, h_colloc.data(), h_colloc.size() * sizeof(ColLoc), hipMemcpyHostToDevice, g->stream));
    if (g->fusable && g->n_groups > 0) {
        std::vector<FusedTile> h_tiles;
        h_tiles.reserve(n_tiles_max);
        for (uint32_t gi = 0; gi < g->n_groups; gi++) {
            const cs_selchunk &sc = s->sel[gi];
            uint32_t rows = r->stripes[sc.stripe].group_rows[sc.chunk];
            uint32_t ntile = (rows + FUSE_TILE_ROWS - 1) / FUSE_TILE_ROWS;
            for (uint32_t k = 0; k < ntile; k++) {
                FusedTile ft{};
                ft.row_count = (k + 1 < ntile) ? FUSE_TILE_ROWS
                                               : rows - k * FUSE_TILE_ROWS;
                for (uint32_t pj = 0; pj < n_proj; pj++)
                    ft.seg_base[pj] = seg_start[(uint64_t)gi * n_proj + pj] + 64 * k;
                h_tiles.push_back(ft);
            }
        }
        g->n_tiles = (uint32_t)h_tiles.size();
        HIP_TRY(hipMalloc(&g->d_tiles, h_tiles.size() * sizeof(FusedTile)));
        HIP_TRY(hipMemcpyAsync(g->d_tiles, h_tiles.data(),
                               h_tiles.size() * sizeof(FusedTile),
                               hipMemcpyHostToDevice, g->stream));
    }
    /* mixed-width (fused grouped) tiles: lane ranges by prefix */
    if (g->fusable_mixed && g->n_groups > 0 && n_proj <= 8) {
        uint32_t lb = 0;
        for (uint32_t pj = 0; pj < n_proj; pj++) {
            /* width from any group's colloc (uniform per column) */
            uint8_t w = h_colloc[pj].width;   /* uniform per column */
            g->fwidth[pj] = w;
            g->flane_base[pj] = (uint16_t)lb;
            lb += (uint32_t)FUSEG_TILE_ROWS * w / 256;
        }
        if (lb <= AGG_BLOCK) {
            g->flanes_total = (uint16_t)lb;
            std::vector<FusedTileG> h_tiles2;
            for (uint32_t gi = 0; gi < g->n_groups; gi++) {
                const cs_selchunk &sc = s->sel[gi];
                uint32_t rows = r->stripes[sc.stripe].group_rows[sc.chunk];
                uint32_t ntile = (rows + FUSEG_TILE_ROWS - 1) / FUSEG_TILE_ROWS;
                for (uint32_t k = 0; k < ntile; k++) {
                    FusedTileG ft{};
                    ft.row_count = (k + 1 < ntile) ? FUSEG_TILE_ROWS
                                                   : rows - k * FUSEG_TILE_ROWS;
                    for (uint32_t pj = 0; pj < n_proj; pj++)
                        ft.seg_base[pj] = seg_start[(uint64_t)gi * n_proj + pj]
                                          + k * ((uint32_t)FUSEG_TILE_ROWS * g->fwidth[pj] / 256);
                    h_tiles2.push_back(ft);
                }
            }
            g->n_tiles2 = (uint32_t)h_tiles2.size();
            HIP_TRY(hipMalloc(&g->d_tiles2, h_tiles2.size() * sizeof(FusedTileG)));
            HIP_TRY(hipMemcpyAsync(g->d_tiles2, h_tiles2.data(),
                                   h_tiles2.size() * sizeof(FusedTileG),
                                   hipMemcpyHostToDevice, g->stream));
        } else {
            g->fusable_mixed = false;
        }
    }
    g->greedy256_mask = 0;
    for (uint32_t pj = 0; pj < n_proj; pj++)
        if (col_g256[pj]) g->greedy256_mask |= 1ull << pj;
    if (g->n_groups > 0 && n_proj > 0) {
        HIP_TRY(hipMalloc(&g->d_segstart, seg_start.size() * 4));
        HIP_TRY(hipMemcpyAsync(g->d_segstart, seg_start.data(),
                               seg_start.size() * 4, hipMemcpyHostToDevice,
                               g->stream));
    }
    HIP_TRY(hipStreamSynchronize(g->stream));
    g->colloc_host = h_colloc;
    return CSTRIPE_OK;
}

/* launch the decode grid: LDS-staged variant when every segment fits the
 * 64 KiB budget (writer default 8 KiB segments -> ~9 waves/CU), else the
 * global-memory fallback */
static int launch_decode(cs_gpu_state *g)
{
    if (g->n_zsegs > 0) {
        /* CSTRIPE_ZR_VARIANT: 0 = force global kernel (A/B knob), default =
         * LDS-staged when the largest frame fits the budget */
        const char *zv = getenv("CSTRIPE_ZR_VARIANT");
        const int force = zv ? atoi(zv) : -1;
        /* stride: largest frame + 8 B slack (the aligned bit window and the
         * funnel loads may read a few bytes past comp_len, into the lane's
         * own slack), 16-aligned; tweaked off bank-aligned lane spacing */
        /* strides in ODD dword counts: lane l's frame starts at l*stride, so
         * equal-progress lanes hit bank (l*stride/4 + c) mod 64 — an odd
         * dword stride makes those 64 distinct banks (even residues conflict
         * 2-32 way; same rule as the lz4 lane kernel's 276/532/1044) */
        uint32_t stride = ((g->max_zseg_comp + 8 + 3) & ~3u) + 4;
        if (((stride / 4) & 1) == 0) stride += 4;
        uint32_t ostride = ((g->max_zseg_dlen + 3) & ~3u) + 4;
        if (((ostride / 4) & 1) == 0) ostride += 4;
        const uint32_t grid = (g->n_zsegs + ZR_LDS_BLOCK - 1) / ZR_LDS_BLOCK;
        const size_t lds1 = (size_t)ZR_LDS_BLOCK * stride + sizeof(zr_dtables);
        const size_t lds2 = lds1 + (size_t)ZR_LDS_BLOCK * ostride;
        if ((force == -1 || force >= 2) && lds2 <= 152 * 1024) {
            hipLaunchKernelGGL(zr_decode_lds2_kernel, dim3(grid), dim3(ZR_LDS_BLOCK),
                               lds2, g->stream,
                               g->d_data, g->d_scratch, g->d_zsegs, g->n_zsegs,
                               stride, ostride, g->d_zrtab, g->d_error);
        } else if (force != 0 && lds1 <= 152 * 1024) {
            hipLaunchKernelGGL(zr_decode_lds_kernel, dim3(grid), dim3(ZR_LDS_BLOCK),
                               lds1, g->stream,
                               g->d_data, g->d_scratch, g->d_zsegs, g->n_zsegs,
                               stride, g->d_zrtab, g->d_error);
        } else {
            const uint32_t grid = (g->n_zsegs + 255) / 256;
            hipLaunchKernelGGL(zr_decode_kernel, dim3(grid), dim3(256), 0, g->stream,
                               g->d_data, g->d_scratch, g->d_zsegs, g->n_zsegs,
                               g->d_zrtab, g->d_error);
        }
        HIP_TRY(hipGetLastError());
    }
    if (g->n_segs == 0) return CSTRIPE_OK;
    /* lane-parallel path for micro-segments (one lane per segment).
     * stride: 16 B-multiple, (stride/4)%64 != 0 so equal-progress lanes land
     * on different banks; LDS/wave = 64*stride -> waves/CU 9 / 4 / 2. */
    if (g->segs_16aligned && g->max_seg_dlen <= 1039) {
        uint32_t block, stride;
        if (g->max_seg_dlen <= 275)      { block = 256; stride = 276; }
        else if (g->max_seg_dlen <= 531) { block = 256; stride = 532; }
        else                             { block = 128; stride = 1044; }
        uint32_t grid = (g->n_segs + block - 1) / block;
        hipLaunchKernelGGL(lz4_decode_lane_kernel, dim3(grid), dim3(block),
                           block * stride + block * 16, g->stream,
                           g->d_data, g->d_scratch, g->d_segs, g->n_segs, stride,
                           g->d_error);
        HIP_TRY(hipGetLastError());
        return CSTRIPE_OK;
    }
    uint32_t in_cap = (g->max_seg_comp + 8 + 15) & ~15u;
    uint32_t out_cap = (g->max_seg_dlen + 15) & ~15u;
    if (in_cap + out_cap <= 64 * 1024) {
        hipLaunchKernelGGL(lz4_decode_lds_kernel, dim3(g->n_segs), dim3(WAVE),
                           in_cap + out_cap, g->stream,
                           g->d_data, g->d_scratch, g->d_segs, in_cap, g->d_error);
    } else {
        hipLaunchKernelGGL(lz4_decode_kernel, dim3(g->n_segs), dim3(WAVE), 0, g->stream,
                           g->d_data, g->d_scratch, g->d_segs, g->d_error);
    }
    /* a failed launch would leave d_scratch stale while later memcpys
       succeed -> silent garbage; surface it as an error (round-1 advisor) */
    HIP_TRY(hipGetLastError());
    return CSTRIPE_OK;
}

/* =====================================================================
 * scan_agg
 * ===================================================================== */

int csgpu_agg(cstripe_scan *s, const cstripe_agg_spec *aggs, uint32_t n_aggs,
              const uint32_t *group_cols, uint32_t n_group_cols,
              cstripe_group_result *gr, cstripe_partial *out)
{
    if (!s->gpu) { cs_set_err("scan not staged — call cstripe_gpu_stage first (GPU required; no CPU fallback)"); return CSTRIPE_ERR_NOGPU; }
    cs_gpu_state *g = s->gpu;
    cstripe_reader *r = s->r;
    if (n_aggs > MAX_AGGS) { cs_set_err("too many aggs"); return CSTRIPE_ERR_ARG; }
    if (s->preds.size() > MAX_PREDS) { cs_set_err("too many preds"); return CSTRIPE_ERR_ARG; }

    AggParams p{};
    p.n_preds = (uint32_t)s->preds.size();
    p.n_aggs = n_aggs;
    p.n_proj = g->n_proj;
    p.tiles_per_group = (r->head.chunk_row_limit + TILE_ROWS - 1) / TILE_ROWS;
    if (p.tiles_per_group == 0) p.tiles_per_group = 1;
    p.n_groups = g->n_groups;
    for (uint32_t i = 0; i < p.n_preds; i++) {
        const cstripe_pred &q = s->preds[i];
        p.preds[i].proj = (uint16_t)g->proj_of_col[q.column];
        p.preds[i].op = (uint8_t)q.op;
        uint8_t t = r->cols[q.column].type;
        p.preds[i].is_float = (t == CSTRIPE_F32 || t == CSTRIPE_F64) ? 1 : 0;
        p.preds[i].gend = (i + 1 == p.n_preds ||
                           s->preds[i + 1].or_group != q.or_group) ? 1 : 0;
        p.preds[i].ival = q.ival;
        p.preds[i].fval = q.fval;
    }
    auto proj_of = [&](int32_t col) -> int {
        if (col < 0 || col >= (int32_t)r->head.column_count) return -1;
        return g->proj_of_col[col];
    };
    for (uint32_t i = 0; i < n_aggs; i++) {
        p.aggs[i].kind = (uint8_t)aggs[i].kind;
        p.aggs[i].one = aggs[i].one;
        int pa = proj_of(aggs[i].col_a), pb = proj_of(aggs[i].col_b), pc = proj_of(aggs[i].col_c);
        if (aggs[i].kind != CSTRIPE_AGG_COUNT_STAR && pa < 0) { cs_set_err("agg %u: col_a not projected", i); return CSTRIPE_ERR_ARG; }
        p.aggs[i].proj_a = (uint8_t)(pa < 0 ? 0 : pa);
        p.aggs[i].proj_b = (uint8_t)(pb < 0 ? 0 : pb);
        p.aggs[i].proj_c = (uint8_t)(pc < 0 ? 0 : pc);
    }

    if (n_group_cols > 0) {
        GroupParams gp{};
        gp.base = p;
        gp.n_group_cols = n_group_cols;
        for (uint32_t i = 0; i < n_group_cols; i++) {
            int pj = proj_of((int32_t)group_cols[i]);
            if (pj < 0) { cs_set_err("group col %u not projected", group_cols[i]); return CSTRIPE_ERR_ARG; }
            if (r->cols[group_cols[i]].type != CSTRIPE_I8 &&
                r->cols[group_cols[i]].type != CSTRIPE_TEXT) { cs_set_err("group col %u must be I8 or TEXT", group_cols[i]); return CSTRIPE_ERR_ARG; }
            gp.gproj[i] = (uint32_t)pj;
        }
        gp.n_work = g->n_groups * p.tiles_per_group;
        if (gp.n_work == 0) {
            gr->n_groups = 0;
            s->last_kernel_ms = s->last_decode_ms = s->last_agg_ms = 0;
            return CSTRIPE_OK;
        }
        const uint32_t n_waves = AGG_BLOCK / WAVE;
        const uint32_t per_block = n_waves * GRP_SLOTS;
        uint32_t grid = gp.n_work < GRP_GRID ? gp.n_work : GRP_GRID;
        if (!g->d_gkeys) {
            HIP_TRY(hipMalloc(&g->d_gkeys, (uint64_t)GRP_GRID * per_block * 4));
            HIP_TRY(hipMalloc(&g->d_gcells, (uint64_t)GRP_GRID * per_block * MAX_AGGS * sizeof(AccCell)));
            HIP_TRY(hipMalloc(&g->d_gfkeys, CSTRIPE_MAX_GROUPS * 4));
            HIP_TRY(hipMalloc(&g->d_gfcells, (uint64_t)CSTRIPE_MAX_GROUPS * MAX_AGGS * sizeof(AccCell)));
            HIP_TRY(hipMalloc(&g->d_gn, 4));
        }
        bool int_aggs_g = true;
        for (uint32_t a = 0; a < n_aggs; a++)
            if (aggs[a].kind == CSTRIPE_AGG_SUM_F64 || aggs[a].kind == CSTRIPE_AGG_MIN_F64 ||
                aggs[a].kind == CSTRIPE_AGG_MAX_F64)
                int_aggs_g = false;
        if (g->all_dense) {
            /* multi-row grouped kernel (windowed loads + per-wave 16-slot
             * LDS tables); falls back below on >16 distinct keys (flag 8).
             * Key columns stored as uniform 256 B greedy-LZ4 segments are
             * decoded INSIDE the kernel (half-tile LDS staging) — when they
             * are the only scratch consumers, the standalone decode kernel
             * and its scratch round trip are skipped entirely. */
            const uint32_t mg_per_block = n_waves * MGRP_SLOTS;
            uint32_t mgrid = gp.n_work < GRP_GRID ? gp.n_work : GRP_GRID;
            gp.kdec_mask = 0;
            uint32_t extra_lds = 0;
            /* fused in-kernel key decode measured SLOWER than the separate
             * massively-parallel decode kernel (idle lanes during the
             * half-tile decode phase + LDS-driven occupancy loss): keep it
             * behind the variant knob for future work */
            static const int gvar2 = [] {
                const char *e = getenv("CSTRIPE_GROUPED_VARIANT");
                return e ? atoi(e) : 0;
            }();
            for (uint32_t i = 0; i < n_group_cols; i++) {
                const uint32_t pj = gp.gproj[i];
                const uint32_t w = g->colloc_host[pj].width;
                gp.kwidth[i] = w;
                if (gvar2 == 2 && ((g->greedy256_mask >> pj) & 1)) {
                    gp.kdec_mask |= 1u << i;
                    extra_lds += ((2048u * w + 255u) / 256u) * 280u;
                }
            }
            uint32_t mlds = (n_waves * MGRP_SLOTS * n_aggs *
                             (uint32_t)sizeof(ThreadAcc) + 15u) & ~15u;
            mlds += extra_lds;
            bool need_decode = g->n_zsegs > 0;
            for (uint32_t pj = 0; pj < g->n_proj && !need_decode; pj++) {
                if (!g->col_scratch[pj]) continue;
                bool fused_key = false;
                for (uint32_t i = 0; i < n_group_cols; i++)
                    fused_key |= gp.gproj[i] == pj && (gp.kdec_mask & (1u << i));
                if (!fused_key) need_decode = true;
            }
            HIP_TRY(hipMemsetAsync(g->d_error, 0, sizeof(int), g->stream));
            HIP_TRY(hipEventRecord(g->ev0, g->stream));
            if (need_decode) { int _rc = launch_decode(g); if (_rc != CSTRIPE_OK) return _rc; }
            HIP_TRY(hipEventRecord(g->ev1, g->stream));
            auto launchmg = [&](auto *kern) {
                hipLaunchKernelGGL(kern, dim3(mgrid), dim3(AGG_BLOCK), mlds, g->stream,
                                   g->d_data, g->d_scratch, g->d_groups, g->d_colloc,
                                   g->d_segs, g->d_segstart,
                                   g->d_gkeys, g->d_gcells, g->d_error, gp);
            };
            static const int gvar = [] {
                const char *e = getenv("CSTRIPE_GROUPED_VARIANT");
                return e ? atoi(e) : 0;
            }();
            if (gvar == 1) {       /* forced 5 waves/SIMD A/B variant */
                if (n_aggs == 5) launchmg(multi_grouped_kernel<5, 8, 5>);
                else if (n_aggs == 1) launchmg(multi_grouped_kernel<1, 8, 5>);
                else if (n_aggs == 2) launchmg(multi_grouped_kernel<2, 8, 5>);
                else if (n_aggs == 4) launchmg(multi_grouped_kernel<4, 8, 5>);
                else launchmg(multi_grouped_kernel<-1, 4>);
            } else {
                if (n_aggs == 5) launchmg(multi_grouped_kernel<5, 8>);
                else if (n_aggs == 1) launchmg(multi_grouped_kernel<1, 8>);
                else if (n_aggs == 2) launchmg(multi_grouped_kernel<2, 8>);
                else if (n_aggs == 4) launchmg(multi_grouped_kernel<4, 8>);
                else launchmg(multi_grouped_kernel<-1, 4>);
            }
            HIP_TRY(hipGetLastError());
            hipLaunchKernelGGL(grouped_final_kernel, dim3(1), dim3(AGG_BLOCK), 0, g->stream,
                               g->d_gkeys, g->d_gcells, mgrid, mg_per_block,
                               g->d_gfkeys, g->d_gfcells, g->d_gn, g->d_error, gp);
            HIP_TRY(hipGetLastError());
            HIP_TRY(hipEventRecord(g->ev2, g->stream));
            int h_err = 0;
            HIP_TRY(hipMemcpyAsync(&h_err, g->d_error, sizeof(int), hipMemcpyDeviceToHost, g->stream));
            HIP_TRY(hipStreamSynchronize(g->stream));
            if (h_err & 4) { cs_set_err("decode error on device (flag %d)", h_err); return CSTRIPE_ERR_FORMAT; }
            if (h_err == 0) {
                s->last_fused = 0;   /* i8 key columns decode to scratch */
                uint32_t h_keys[CSTRIPE_MAX_GROUPS];
                std::vector<AccCell> h_cells((size_t)CSTRIPE_MAX_GROUPS * n_aggs);
                uint32_t h_n = 0;
                HIP_TRY(hipMemcpyAsync(h_keys, g->d_gfkeys, sizeof(h_keys), hipMemcpyDeviceToHost, g->stream));
                HIP_TRY(hipMemcpyAsync(h_cells.data(), g->d_gfcells, h_cells.size() * sizeof(AccCell), hipMemcpyDeviceToHost, g->stream));
                HIP_TRY(hipMemcpyAsync(&h_n, g->d_gn, 4, hipMemcpyDeviceToHost, g->stream));
                HIP_TRY(hipStreamSynchronize(g->stream));
                float ms_d = 0, ms_k = 0;
                (void)hipEventElapsedTime(&ms_d, g->ev0, g->ev1);
                (void)hipEventElapsedTime(&ms_k, g->ev1, g->ev2);
                s->last_decode_ms = ms_d;
                s->last_agg_ms = ms_k;
                s->last_kernel_ms = ms_d + ms_k;
                std::vector<uint32_t> order(h_n);
                for (uint32_t i = 0; i < h_n; i++) order[i] = i;
                std::sort(order.begin(), order.end(),
                          [&](uint32_t a, uint32_t b) { return h_keys[a] < h_keys[b]; });
                gr->n_groups = h_n;
                for (uint32_t oi = 0; oi < h_n; oi++) {
                    uint32_t i = order[oi];
                    gr->keys[oi] = h_keys[i];
                    for (uint32_t a = 0; a < n_aggs; a++) {
                        const AccCell &cc2 = h_cells[(size_t)i * n_aggs + a];
                        cstripe_partial o{};
                        o.count = cc2.cnt;
                        o.is_null = (cc2.cnt == 0) ? 1 : 0;
                        switch (aggs[a].kind) {
                            case CSTRIPE_AGG_COUNT_STAR:
                            case CSTRIPE_AGG_COUNT_COL:
                                o.i128_lo = cc2.cnt; o.is_null = 0; break;
                            case CSTRIPE_AGG_SUM_F64:
                            case CSTRIPE_AGG_MIN_F64:
                            case CSTRIPE_AGG_MAX_F64:
                                if (!o.is_null) o.f64 = cc2.f; break;
                            case CSTRIPE_AGG_MIN_I64:
                            case CSTRIPE_AGG_MAX_I64:
                                if (!o.is_null) { o.i128_lo = cc2.lo; o.i128_hi = cc2.lo < 0 ? -1 : 0; }
                                break;
                            default:
                                if (!o.is_null) { o.i128_lo = cc2.lo; o.i128_hi = cc2.hi; }
                                break;
                        }
                        out[(size_t)oi * n_aggs + a] = o;
                    }
                }
                return CSTRIPE_OK;
            }
            /* flag 8: >16 distinct keys — fall through to the 64-group path */
        }
        if (g->fusable_mixed && int_aggs_g && g->d_tiles2) {
            /* fused grouped: decode tile to LDS + atomic group table; falls
             * back below when >16 distinct groups (device flag 8) */
            FusedGParams fgp{};
            fgp.base = p;
            fgp.n_group_cols = n_group_cols;
            for (uint32_t i = 0; i < n_group_cols; i++) fgp.gproj[i] = gp.gproj[i];
            fgp.n_tiles = g->n_tiles2;
            for (uint32_t i = 0; i < 8; i++) { fgp.lane_base[i] = g->flane_base[i]; fgp.width[i] = g->fwidth[i]; }
            fgp.lanes_total = g->flanes_total;
            uint32_t fgrid = g->n_tiles2 < FUSEG_GRID ? g->n_tiles2 : FUSEG_GRID;
            uint32_t flds = (uint32_t)fgp.lanes_total * FUSE_STRIDE + FUSEG_SLOTS * 4
                            + FUSEG_SLOTS * n_aggs * (uint32_t)sizeof(ThreadAcc);
            HIP_TRY(hipMemsetAsync(g->d_error, 0, sizeof(int), g->stream));
            HIP_TRY(hipEventRecord(g->ev0, g->stream));
            auto launchfg = [&](auto *kern) {
                hipLaunchKernelGGL(kern, dim3(fgrid), dim3(AGG_BLOCK), flds, g->stream,
                                   g->d_data, g->d_segs, g->d_tiles2, g->d_gkeys,
                                   g->d_gcells, g->d_error, fgp);
            };
            if (n_aggs == 5) launchfg(fused_grouped_kernel<5>);
            else if (n_aggs == 1) launchfg(fused_grouped_kernel<1>);
            else if (n_aggs == 2) launchfg(fused_grouped_kernel<2>);
            else launchfg(fused_grouped_kernel<-1>);
            HIP_TRY(hipGetLastError());
            HIP_TRY(hipEventRecord(g->ev1, g->stream));
            hipLaunchKernelGGL(grouped_final_kernel, dim3(1), dim3(AGG_BLOCK), 0, g->stream,
                               g->d_gkeys, g->d_gcells, fgrid, FUSEG_SLOTS,
                               g->d_gfkeys, g->d_gfcells, g->d_gn, g->d_error, gp);
        HIP_TRY(hipGetLastError());
            HIP_TRY(hipEventRecord(g->ev2, g->stream));
            int h_err = 0;
            HIP_TRY(hipMemcpyAsync(&h_err, g->d_error, sizeof(int), hipMemcpyDeviceToHost, g->stream));
            HIP_TRY(hipStreamSynchronize(g->stream));
            if (h_err & 4) { cs_set_err("decode error on device (flag %d)", h_err); return CSTRIPE_ERR_FORMAT; }
            if (h_err == 0) {
                s->last_fused = 1;
                uint32_t h_keys[CSTRIPE_MAX_GROUPS];
                std::vector<AccCell> h_cells((size_t)CSTRIPE_MAX_GROUPS * n_aggs);
                uint32_t h_n = 0;
                HIP_TRY(hipMemcpyAsync(h_keys, g->d_gfkeys, sizeof(h_keys), hipMemcpyDeviceToHost, g->stream));
                HIP_TRY(hipMemcpyAsync(h_cells.data(), g->d_gfcells, h_cells.size() * sizeof(AccCell), hipMemcpyDeviceToHost, g->stream));
                HIP_TRY(hipMemcpyAsync(&h_n, g->d_gn, 4, hipMemcpyDeviceToHost, g->stream));
                HIP_TRY(hipStreamSynchronize(g->stream));
                float ms_f = 0, ms_r = 0;
                (void)hipEventElapsedTime(&ms_f, g->ev0, g->ev1);
                (void)hipEventElapsedTime(&ms_r, g->ev1, g->ev2);
                s->last_decode_ms = ms_f;
                s->last_agg_ms = ms_r;
                s->last_kernel_ms = ms_f + ms_r;
                std::vector<uint32_t> order(h_n);
                for (uint32_t i = 0; i < h_n; i++) order[i] = i;
                std::sort(order.begin(), order.end(),
                          [&](uint32_t a, uint32_t b) { return h_keys[a] < h_keys[b]; });
                gr->n_groups = h_n;
                for (uint32_t oi = 0; oi < h_n; oi++) {
                    uint32_t i = order[oi];
                    gr->keys[oi] = h_keys[i];
                    for (uint32_t a = 0; a < n_aggs; a++) {
                        const AccCell &cc2 = h_cells[(size_t)i * n_aggs + a];
                        cstripe_partial o{};
                        o.count = cc2.cnt;
                        o.is_null = (cc2.cnt == 0) ? 1 : 0;
                        switch (aggs[a].kind) {
                            case CSTRIPE_AGG_COUNT_STAR:
                            case CSTRIPE_AGG_COUNT_COL:
                                o.i128_lo = cc2.cnt; o.is_null = 0; break;
                            case CSTRIPE_AGG_MIN_I64:
                            case CSTRIPE_AGG_MAX_I64:
                                if (!o.is_null) { o.i128_lo = cc2.lo; o.i128_hi = cc2.lo < 0 ? -1 : 0; }
                                break;
                            default:
                                if (!o.is_null) { o.i128_lo = cc2.lo; o.i128_hi = cc2.hi; }
                                break;
                        }
                        out[(size_t)oi * n_aggs + a] = o;
                    }
                }
                return CSTRIPE_OK;
            }
        }

        uint32_t lds = n_waves * GRP_SLOTS * 4 + n_waves * GRP_SLOTS * n_aggs * (uint32_t)sizeof(ThreadAcc);
        HIP_TRY(hipMemsetAsync(g->d_error, 0, sizeof(int), g->stream));
        HIP_TRY(hipEventRecord(g->ev0, g->stream));
        { int _rc = launch_decode(g); if (_rc != CSTRIPE_OK) return _rc; }
        HIP_TRY(hipEventRecord(g->ev1, g->stream));
        {
            auto launchg = [&](auto *kern) {
                hipLaunchKernelGGL(kern, dim3(grid), dim3(AGG_BLOCK), lds, g->stream,
                                   g->d_data, g->d_scratch, g->d_rank, g->d_groups,
                                   g->d_colloc, g->d_gkeys, g->d_gcells, g->d_error, gp);
            };
            if (n_aggs == 5) launchg(grouped_agg_kernel<5>);
            else if (n_aggs == 1) launchg(grouped_agg_kernel<1>);
            else if (n_aggs == 2) launchg(grouped_agg_kernel<2>);
            else launchg(grouped_agg_kernel<-1>);
            HIP_TRY(hipGetLastError());
        }
        hipLaunchKernelGGL(grouped_final_kernel, dim3(1), dim3(AGG_BLOCK), 0, g->stream,
                           g->d_gkeys, g->d_gcells, grid, per_block,
                           g->d_gfkeys, g->d_gfcells, g->d_gn, g->d_error, gp);
        HIP_TRY(hipGetLastError());
        HIP_TRY(hipEventRecord(g->ev2, g->stream));

        uint32_t h_keys[CSTRIPE_MAX_GROUPS];
        std::vector<AccCell> h_cells((size_t)CSTRIPE_MAX_GROUPS * n_aggs);
        uint32_t h_n = 0;
        int h_err = 0;
        HIP_TRY(hipMemcpyAsync(h_keys, g->d_gfkeys, sizeof(h_keys), hipMemcpyDeviceToHost, g->stream));
        HIP_TRY(hipMemcpyAsync(h_cells.data(), g->d_gfcells, h_cells.size() * sizeof(AccCell), hipMemcpyDeviceToHost, g->stream));
        HIP_TRY(hipMemcpyAsync(&h_n, g->d_gn, 4, hipMemcpyDeviceToHost, g->stream));
        HIP_TRY(hipMemcpyAsync(&h_err, g->d_error, sizeof(int), hipMemcpyDeviceToHost, g->stream));
        HIP_TRY(hipStreamSynchronize(g->stream));
        if (h_err & 4) { cs_set_err("decode error on device (flag %d)", h_err); return CSTRIPE_ERR_FORMAT; }
        if (h_err) { cs_set_err("too many distinct groups (device flag %d; caps: %d/wave, %d total)", h_err, GRP_SLOTS, CSTRIPE_MAX_GROUPS); return CSTRIPE_ERR; }

        float ms_decode = 0, ms_agg = 0;
        (void)hipEventElapsedTime(&ms_decode, g->ev0, g->ev1);
        (void)hipEventElapsedTime(&ms_agg, g->ev1, g->ev2);
        s->last_decode_ms = ms_decode;
        s->last_agg_ms = ms_agg;
        s->last_kernel_ms = ms_decode + ms_agg;

        /* sort groups by key for deterministic output (oracle does the same) */
        std::vector<uint32_t> order(h_n);
        for (uint32_t i = 0; i < h_n; i++) order[i] = i;
        std::sort(order.begin(), order.end(),
                  [&](uint32_t a, uint32_t b) { return h_keys[a] < h_keys[b]; });
        gr->n_groups = h_n;
        for (uint32_t oi = 0; oi < h_n; oi++) {
            uint32_t i = order[oi];
            gr->keys[oi] = h_keys[i];
            for (uint32_t a = 0; a < n_aggs; a++) {
                const AccCell &c = h_cells[(size_t)i * n_aggs + a];
                cstripe_partial o{};
                o.count = c.cnt;
                o.is_null = (c.cnt == 0) ? 1 : 0;
                switch (aggs[a].kind) {
                    case CSTRIPE_AGG_COUNT_STAR:
                    case CSTRIPE_AGG_COUNT_COL:
                        o.i128_lo = c.cnt; o.is_null = 0; break;
                    case CSTRIPE_AGG_SUM_F64:
                    case CSTRIPE_AGG_MIN_F64:
                    case CSTRIPE_AGG_MAX_F64:
                        if (!o.is_null) o.f64 = c.f; break;
                    case CSTRIPE_AGG_MIN_I64:
                    case CSTRIPE_AGG_MAX_I64:
                        if (!o.is_null) { o.i128_lo = c.lo; o.i128_hi = c.lo < 0 ? -1 : 0; } break;
                    default:
                        if (!o.is_null) { o.i128_lo = c.lo; o.i128_hi = c.hi; } break;
                }
                out[(size_t)oi * n_aggs + a] = o;
            }
        }
        return CSTRIPE_OK;
    }

    uint32_t n_blocks = g->n_groups * p.tiles_per_group;
    if (n_blocks == 0) {
        /* empty selection — all-NULL partials (combine applies COUNT coalesce) */
        for (uint32_t a = 0; a < n_aggs; a++) {
            cstripe_partial z{};
            z.is_null = 1;
            if (aggs[a].kind == CSTRIPE_AGG_COUNT_STAR || aggs[a].kind == CSTRIPE_AGG_COUNT_COL) z.is_null = 0;
            out[a] = z;
        }
        s->last_kernel_ms = s->last_decode_ms = s->last_agg_ms = 0;
        return CSTRIPE_OK;
    }

    bool int_aggs = true;
    for (uint32_t a = 0; a < n_aggs; a++)
        if (aggs[a].kind == CSTRIPE_AGG_SUM_F64 || aggs[a].kind == CSTRIPE_AGG_MIN_F64 ||
            aggs[a].kind == CSTRIPE_AGG_MAX_F64)
            int_aggs = false;

    HIP_TRY(hipMemsetAsync(g->d_error, 0, sizeof(int), g->stream));
    if (g->fusable && int_aggs && g->d_tiles) {
        /* fused decode+filter+aggregate: no scratch round trip */
        s->last_fused = 1;
        HIP_TRY(hipEventRecord(g->ev0, g->stream));
        auto launchf = [&](auto *kern) {
            hipLaunchKernelGGL(kern, dim3(g->n_tiles), dim3(AGG_BLOCK),
                               AGG_BLOCK * FUSE_STRIDE, g->stream,
                               g->d_data, g->d_segs, g->d_tiles, g->d_block,
                               g->d_error, p);
        };
        if (p.n_preds == 5 && n_aggs == 2) launchf(fused_agg_kernel<5, 2>);
        else if (p.n_preds == 5 && n_aggs == 4) launchf(fused_agg_kernel<5, 4>);
        else if (p.n_preds == 5 && n_aggs == 1) launchf(fused_agg_kernel<5, 1>);
        else if (p.n_preds == 1 && n_aggs == 1) launchf(fused_agg_kernel<1, 1>);
        else if (p.n_preds == 2 && n_aggs == 2) launchf(fused_agg_kernel<2, 2>);
        else launchf(fused_agg_kernel<-1, -1>);
        HIP_TRY(hipGetLastError());
        HIP_TRY(hipEventRecord(g->ev1, g->stream));
        { int _rc = launch_final_reduce(g, g->n_tiles, p); if (_rc != CSTRIPE_OK) return _rc; }
        HIP_TRY(hipEventRecord(g->ev2, g->stream));

        AccCell h_final[MAX_AGGS];
        int h_err = 0;
        HIP_TRY(hipMemcpyAsync(h_final, g->d_final, n_aggs * sizeof(AccCell), hipMemcpyDeviceToHost, g->stream));
        HIP_TRY(hipMemcpyAsync(&h_err, g->d_error, sizeof(int), hipMemcpyDeviceToHost, g->stream));
        HIP_TRY(hipStreamSynchronize(g->stream));
        if (h_err) { cs_set_err("decode error on device (flag %d)", h_err); return CSTRIPE_ERR_FORMAT; }

        float ms_fused = 0, ms_red = 0;
        (void)hipEventElapsedTime(&ms_fused, g->ev0, g->ev1);
        (void)hipEventElapsedTime(&ms_red, g->ev1, g->ev2);
        s->last_decode_ms = ms_fused;     /* the fused kernel (decode dominates) */
        s->last_agg_ms = ms_red;
        s->last_kernel_ms = ms_fused + ms_red;

        for (uint32_t a = 0; a < n_aggs; a++) {
            cstripe_partial o{};
            const AccCell &cc = h_final[a];
            o.count = cc.cnt;
            o.is_null = (cc.cnt == 0) ? 1 : 0;
            if (o.is_null && aggs[a].kind != CSTRIPE_AGG_COUNT_STAR &&
                aggs[a].kind != CSTRIPE_AGG_COUNT_COL) { out[a] = o; continue; }
            switch (aggs[a].kind) {
                case CSTRIPE_AGG_COUNT_STAR:
                case CSTRIPE_AGG_COUNT_COL:
                    o.i128_lo = cc.cnt; o.is_null = 0; break;
                case CSTRIPE_AGG_MIN_I64:
                case CSTRIPE_AGG_MAX_I64:
                    o.i128_lo = cc.lo; o.i128_hi = cc.lo < 0 ? -1 : 0; break;
                default:
                    o.i128_lo = cc.lo; o.i128_hi = cc.hi; break;
            }
            out[a] = o;
        }
        return CSTRIPE_OK;
    }

    s->last_fused = 0;
    HIP_TRY(hipEventRecord(g->ev0, g->stream));
    { int _rc = launch_decode(g); if (_rc != CSTRIPE_OK) return _rc; }
    HIP_TRY(hipEventRecord(g->ev1, g->stream));
    if (g->all_dense) {
        /* all-dense: paired-row kernel — one wide load per column covers
         * two rows (canonical closed-form or decoded/raw typed arrays) */
        auto launchp = [&](auto *kern) {
            hipLaunchKernelGGL(kern, dim3(n_blocks), dim3(AGG_BLOCK), 0, g->stream,
                               g->d_data, g->d_scratch, g->d_groups,
                               g->d_colloc, g->d_block, p);
        };
        /* variant knob for on-hardware A/B (default = R8, natural occupancy) */
        static const int kvar = [] {
            const char *e = getenv("CSTRIPE_KERNEL_VARIANT");
            return e ? atoi(e) : 1;     /* measured best: R8 + 8 waves/SIMD */
        }();
        if (kvar == 3 && g->all_canonP) {
            /* experimental LDS-staged scan: bulk-load pred streams to LDS */
            uint32_t tiles2 = (r->head.chunk_row_limit + LDSK_TILE - 1) / LDSK_TILE;
            if (tiles2 == 0) tiles2 = 1;
            const uint32_t nb2 = g->n_groups * tiles2;
            uint32_t lds_need = 0;
            bool lds_ok = true;
            {
                bool seen[MAX_PROJ] = {};
                for (uint32_t i = 0; i < p.n_preds; i++) {
                    const uint32_t pj = p.preds[i].proj;
                    if (seen[pj]) continue;
                    seen[pj] = true;
                    uint32_t maxstep = 0;
                    for (uint32_t gi = 0; gi < g->n_groups; gi++) {
                        const ColLoc &cl = g->colloc_host[(uint64_t)gi * g->n_proj + pj];
                        if (cl.mode != CSF_SEGMODE_CONST)
                            maxstep = max(maxstep, (uint32_t)cl.L + 3u);
                    }
                    lds_need += LDSK_TILE * maxstep + 64;
                }
                if (lds_need > 100 * 1024 || p.n_preds == 0) lds_ok = false;
            }
            if (lds_ok) {
                auto launchl = [&](auto *kern) {
                    hipLaunchKernelGGL(kern, dim3(nb2), dim3(AGG_BLOCK), lds_need,
                                       g->stream, g->d_data, g->d_scratch,
                                       g->d_groups, g->d_colloc, g->d_block,
                                       tiles2, p);
                };
                if (p.n_preds == 5 && n_aggs == 2) launchl(lds_agg_kernel<5, 2>);
                else if (p.n_preds == 5 && n_aggs == 1) launchl(lds_agg_kernel<5, 1>);
                else if (p.n_preds == 1 && n_aggs == 1) launchl(lds_agg_kernel<1, 1>);
                else launchl(lds_agg_kernel<-1, -1>);
                HIP_TRY(hipGetLastError());
                HIP_TRY(hipEventRecord(g->ev1, g->stream));
                { int _rc = launch_final_reduce(g, nb2, p); if (_rc != CSTRIPE_OK) return _rc; }
                HIP_TRY(hipEventRecord(g->ev2, g->stream));
                goto collect_ungrouped;
            }
        }
        if (kvar == 1) {        /* forced 8 waves/SIMD (64 VGPRs, some spill) */
            if (p.n_preds == 5 && n_aggs == 2) launchp(multi_agg_kernel<5, 2, 8, 8>);
            else if (p.n_preds == 5 && n_aggs == 1) launchp(multi_agg_kernel<5, 1, 8, 8>);
            else if (p.n_preds == 1 && n_aggs == 1) launchp(multi_agg_kernel<1, 1, 8, 8>);
            else if (p.n_preds == 2 && n_aggs == 2) launchp(multi_agg_kernel<2, 2, 8, 8>);
            else if (p.n_preds == 5 && n_aggs == 4) launchp(multi_agg_kernel<5, 4, 8, 8>);
            else launchp(multi_agg_kernel<-1, -1, 4>);
        } else if (kvar == 4) { /* R8 + 7 waves/SIMD */
            if (p.n_preds == 5 && n_aggs == 2) launchp(multi_agg_kernel<5, 2, 8, 7>);
            else if (p.n_preds == 5 && n_aggs == 1) launchp(multi_agg_kernel<5, 1, 8, 7>);
            else if (p.n_preds == 1 && n_aggs == 1) launchp(multi_agg_kernel<1, 1, 8, 7>);
            else launchp(multi_agg_kernel<-1, -1, 4>);
        } else if (kvar == 2) { /* R=4 */
            if (p.n_preds == 5 && n_aggs == 2) launchp(multi_agg_kernel<5, 2, 4>);
            else if (p.n_preds == 5 && n_aggs == 1) launchp(multi_agg_kernel<5, 1, 4>);
            else if (p.n_preds == 1 && n_aggs == 1) launchp(multi_agg_kernel<1, 1, 4>);
            else if (p.n_preds == 2 && n_aggs == 2) launchp(multi_agg_kernel<2, 2, 4>);
            else if (p.n_preds == 5 && n_aggs == 4) launchp(multi_agg_kernel<5, 4, 4>);
            else launchp(multi_agg_kernel<-1, -1, 4>);
        } else {
            if (p.n_preds == 5 && n_aggs == 2) launchp(multi_agg_kernel<5, 2, 8>);
            else if (p.n_preds == 5 && n_aggs == 1) launchp(multi_agg_kernel<5, 1, 8>);
            else if (p.n_preds == 1 && n_aggs == 1) launchp(multi_agg_kernel<1, 1, 8>);
            else if (p.n_preds == 2 && n_aggs == 2) launchp(multi_agg_kernel<2, 2, 8>);
            else if (p.n_preds == 5 && n_aggs == 4) launchp(multi_agg_kernel<5, 4, 8>);
            else launchp(multi_agg_kernel<-1, -1, 4>);
        }
        HIP_TRY(hipGetLastError());
    } else {
        auto launch = [&](auto *kern) {
            hipLaunchKernelGGL(kern, dim3(n_blocks), dim3(AGG_BLOCK), 0, g->stream,
                               g->d_data, g->d_scratch, g->d_rank, g->d_groups,
                               g->d_colloc, g->d_block, p);
        };
        /* hot shapes compiled with unrolled pred/agg loops */
        if (p.n_preds == 5 && n_aggs == 2) launch(filter_agg_kernel<5, 2>);
        else if (p.n_preds == 5 && n_aggs == 1) launch(filter_agg_kernel<5, 1>);
        else if (p.n_preds == 1 && n_aggs == 1) launch(filter_agg_kernel<1, 1>);
        else if (p.n_preds == 2 && n_aggs == 2) launch(filter_agg_kernel<2, 2>);
        else launch(filter_agg_kernel<-1, -1>);
        HIP_TRY(hipGetLastError());
    }
    { int _rc = launch_final_reduce(g, n_blocks, p); if (_rc != CSTRIPE_OK) return _rc; }
    HIP_TRY(hipEventRecord(g->ev2, g->stream));

collect_ungrouped:
    ;
    AccCell h_final[MAX_AGGS];
    int h_err = 0;
    HIP_TRY(hipMemcpyAsync(h_final, g->d_final, n_aggs * sizeof(AccCell), hipMemcpyDeviceToHost, g->stream));
    HIP_TRY(hipMemcpyAsync(&h_err, g->d_error, sizeof(int), hipMemcpyDeviceToHost, g->stream));
    HIP_TRY(hipStreamSynchronize(g->stream));
    if (h_err) { cs_set_err("decode error on device (flag %d)", h_err); return CSTRIPE_ERR_FORMAT; }

    float ms_decode = 0, ms_agg = 0;
    (void)hipEventElapsedTime(&ms_decode, g->ev0, g->ev1);
    (void)hipEventElapsedTime(&ms_agg, g->ev1, g->ev2);
    s->last_decode_ms = ms_decode;
    s->last_agg_ms = ms_agg;
    s->last_kernel_ms = ms_decode + ms_agg;

    for (uint32_t a = 0; a < n_aggs; a++) {
        cstripe_partial o{};
        const AccCell &c = h_final[a];
        o.count = c.cnt;
        o.is_null = (c.cnt == 0) ? 1 : 0;
        if (o.is_null && aggs[a].kind != CSTRIPE_AGG_COUNT_STAR &&
            aggs[a].kind != CSTRIPE_AGG_COUNT_COL) { out[a] = o; continue; }
        switch (aggs[a].kind) {
            case CSTRIPE_AGG_COUNT_STAR:
            case CSTRIPE_AGG_COUNT_COL:
                o.count = c.cnt;
                o.i128_lo = c.cnt;
                o.is_null = 0;
                break;
            case CSTRIPE_AGG_SUM_F64:
            case CSTRIPE_AGG_MIN_F64:
            case CSTRIPE_AGG_MAX_F64:
                o.f64 = c.f;
                break;
            case CSTRIPE_AGG_MIN_I64:
            case CSTRIPE_AGG_MAX_I64:
                o.i128_lo = c.lo;
                o.i128_hi = c.lo < 0 ? -1 : 0;
                break;
            default:
                o.i128_lo = c.lo;
                o.i128_hi = c.hi;
                break;
        }
        out[a] = o;
    }
    return CSTRIPE_OK;
}

/* =====================================================================
 * next_batch — parity/fallback access: GPU-decode the next surviving chunk
 * group, copy decoded value streams back, expand to row-aligned on host
 * using the exists bitmap (ReadChunkGroupNextRow contract,
 * columnar_reader.c:868-901, batched).
 * ===================================================================== */

int csgpu_fetch_batch(cstripe_scan *s, uint32_t gi, cstripe_batch *batch)
{
    if (!s->gpu) { cs_set_err("scan not staged — call cstripe_gpu_stage first"); return CSTRIPE_ERR_NOGPU; }
    cs_gpu_state *g = s->gpu;
    cstripe_reader *r = s->r;
    if (gi >= s->sel.size()) return CSTRIPE_END;

    /* make sure scratch holds decoded data (decode everything once) */
    if ((g->n_segs > 0 || g->n_zsegs > 0) && !g->batch_decoded) {
        HIP_TRY(hipMemsetAsync(g->d_error, 0, sizeof(int), g->stream));
        { int _rc = launch_decode(g); if (_rc != CSTRIPE_OK) return _rc; }
        int h_err = 0;
        HIP_TRY(hipMemcpyAsync(&h_err, g->d_error, sizeof(int), hipMemcpyDeviceToHost, g->stream));
        HIP_TRY(hipStreamSynchronize(g->stream));
        if (h_err) { cs_set_err("decode error on device (flag %d)", h_err); return CSTRIPE_ERR_FORMAT; }
        g->batch_decoded = true;
    }

    const cs_selchunk &sc = s->sel[gi];
    const cs_stripe_info &st = r->stripes[sc.stripe];
    uint32_t rows = st.group_rows[sc.chunk];
    const uint8_t *stripe_base = r->stripe_base(st);

    std::vector<uint8_t> packed;
    for (uint32_t c = 0; c < r->head.column_count; c++) {
        int pj = g->proj_of_col[c];
        if (batch->col_values && batch->col_values[c] == nullptr) continue;
        if (pj < 0) continue;
        const cs_skipnode &nd = st.nodes[c][sc.chunk];
        uint32_t width = csf_type_width(r->cols[c].type);
        uint64_t soff = g->scratch_off[(uint64_t)gi * g->n_proj + pj];

        packed.resize(nd.n.decompressed_size);
        const ColLoc &cl = g->colloc_host[(uint64_t)gi * g->n_proj + pj];
        if (cl.flags & 4) {
            /* canonical chunk: closed-form device decode into the temp
             * buffer, then copy back (the agg path never materializes it) */
            if (g->tmp_bytes < nd.n.decompressed_size) {
                if (g->d_tmp) HIP_TRY(hipFree(g->d_tmp));
                g->tmp_bytes = align_up(nd.n.decompressed_size, 4096);
                HIP_TRY(hipMalloc(&g->d_tmp, g->tmp_bytes));
            }
            const uint32_t nv = (uint32_t)(nd.n.decompressed_size / width);
            const uint32_t grid2 = (nv + 255) / 256;
            hipLaunchKernelGGL(canon_decode_kernel, dim3(grid2), dim3(256), 0, g->stream,
                               g->d_data + cl.val_off, g->d_tmp, nv,
                               (uint32_t)cl.mode, (uint32_t)cl.L, width,
                               (uint64_t)cl.hval);
            HIP_TRY(hipGetLastError());
            HIP_TRY(hipMemcpyAsync(packed.data(), g->d_tmp, nd.n.decompressed_size,
                                   hipMemcpyDeviceToHost, g->stream));
            HIP_TRY(hipStreamSynchronize(g->stream));
        } else if (soff != ~0ull) {  /* LZ4-decoded on device */
            HIP_TRY(hipMemcpyAsync(packed.data(), g->d_scratch + soff, nd.n.decompressed_size,
                                   hipMemcpyDeviceToHost, g->stream));
            HIP_TRY(hipStreamSynchronize(g->stream));
        } else {
            /* NONE / host-predecoded zstd: take staged raw from file side */
            if (nd.n.comp_type == CSTRIPE_COMP_NONE) {
                memcpy(packed.data(), stripe_base + nd.n.value_off, nd.n.decompressed_size);
            } else { /* zstd / pglz: decode host-side as at stage */
                for (const csf_seg &sg : st.nodes[c][sc.chunk].segs) {
                    if (nd.n.comp_type == CSTRIPE_COMP_ZSTD) {
                        size_t zr = ZSTD_decompress(packed.data() + sg.decomp_off, sg.decomp_len,
                                                    stripe_base + nd.n.value_off + sg.comp_off, sg.comp_len);
                        if (ZSTD_isError(zr) || zr != sg.decomp_len) { cs_set_err("zstd host decode failed"); return CSTRIPE_ERR_FORMAT; }
                    } else {
                        const uint8_t *pb = stripe_base + nd.n.value_off + sg.comp_off;
                        if (sg.comp_len < CSPGLZ_HDRSZ ||
                            cspglz_decompress(pb + CSPGLZ_HDRSZ,
                                              (int32_t)(sg.comp_len - CSPGLZ_HDRSZ),
                                              packed.data() + sg.decomp_off,
                                              (int32_t)sg.decomp_len) < 0) {
                            cs_set_err("pglz host decode failed");
                            return CSTRIPE_ERR_FORMAT;
                        }
                    }
                }
            }
        }

        /* expand packed -> row-aligned using the exists bitmap */
        uint8_t *dstv = (uint8_t *)batch->col_values[c];
        uint8_t *dstn = batch->col_nulls ? batch->col_nulls[c] : nullptr;
        const uint8_t *eb = stripe_base + nd.n.exists_off;
        uint32_t vidx = 0;
        if (nd.n.n_present == nd.n.row_count) {
            memcpy(dstv, packed.data(), (size_t)rows * width);
            if (dstn) memset(dstn, 0, rows);
        } else {
            for (uint32_t i = 0; i < rows; i++) {
                bool present = (eb[i / 8] >> (i % 8)) & 1;
                if (present) {
                    memcpy(dstv + (size_t)i * width, packed.data() + (size_t)vidx * width, width);
                    vidx++;
                } else {
                    memset(dstv + (size_t)i * width, 0, width);
                }
                if (dstn) dstn[i] = present ? 0 : 1;
            }
        }
    }

    batch->n_rows = rows;
    uint64_t first = st.meta.first_row_number;
    for (uint32_t k = 0; k < sc.chunk; k++) first += st.group_rows[k];
    batch->first_row_number = first;
    return CSTRIPE_OK;
}

int csgpu_next_batch(cstripe_scan *s, cstripe_batch *batch)
{
    int rc = csgpu_fetch_batch(s, (uint32_t)s->batch_pos, batch);
    if (rc == CSTRIPE_OK) s->batch_pos++;
    return rc;
}

/* =====================================================================
 * Device write path: compress one chunk's columns ON the
 * GPU. Stats kernel reduces min/max and the OR-of-XOR signature that
 * decides the canonical mode; the emit kernel then writes the canonical
 * P(L)/CONST stream with one lane per value (every byte position is
 * closed-form — the same property the read path exploits). Columns that
 * do not fit a canonical parse are copied back raw and compressed by the
 * host writer path (greedy LZ4 / zstd / pglz), so any data works.
 * Replaces, for HBM-resident data, the host serializer of
 * SerializeChunkData/CompressBuffer (columnar_writer.c:592-654,
 * columnar_compression.c:62-158); only compressed bytes cross PCIe.
 * ===================================================================== */

struct WStats {
    int64_t mn, mx;            /* int: values; float: f64 bit patterns */
    uint64_t xsig;             /* OR of (v ^ v0) over the chunk (width 8) */
};

template <typename T>
__global__ __launch_bounds__(AGG_BLOCK) void wstats_kernel(
    const T *__restrict__ v, uint32_t n, int is_float, WStats *__restrict__ out)
{
    const uint32_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t xs = 0;
    uint64_t v0 = 0;
    if (n) __builtin_memcpy(&v0, &v[0], sizeof(T) < 8 ? sizeof(T) : 8);
    double fmn = 0, fmx = 0;
    bool seen = false;
    int64_t imn = INT64_MAX, imx = INT64_MIN;
    for (uint32_t i = tid; i < n; i += gridDim.x * blockDim.x) {
        if (is_float) {
            double d;
            if (sizeof(T) == 4) { float f; __builtin_memcpy(&f, &v[i], 4); d = f; }
            else __builtin_memcpy(&d, &v[i], 8);
            if (!seen) { fmn = fmx = d; seen = true; }
            else {
                if (f64cmp_pg(d, fmn) < 0) fmn = d;
                if (f64cmp_pg(d, fmx) > 0) fmx = d;
            }
        } else {
            const int64_t x = (int64_t)v[i];
            imn = min(imn, x);
            imx = max(imx, x);
        }
        if (sizeof(T) == 8) {               /* canonical signature: BIT xor */
            uint64_t bits;
            __builtin_memcpy(&bits, &v[i], 8);
            xs |= bits ^ v0;
        }
    }
    /* wave + LDS + global-atomic reduce */
    __shared__ int64_t smn[AGG_BLOCK / WAVE], smx[AGG_BLOCK / WAVE];
    __shared__ uint64_t sxs[AGG_BLOCK / WAVE];
    __shared__ int sseen[AGG_BLOCK / WAVE];
    for (int d = WAVE / 2; d > 0; d >>= 1) {
        const int64_t omn = __shfl_down((long long)(is_float ? (int64_t)__double_as_longlong(fmn) : imn), d, WAVE);
        const int64_t omx = __shfl_down((long long)(is_float ? (int64_t)__double_as_longlong(fmx) : imx), d, WAVE);
        const uint64_t oxs = (uint64_t)__shfl_down((long long)xs, d, WAVE);
        const int osn = __shfl_down((int)seen, d, WAVE);
        if (is_float) {
            double a, b;
            a = __longlong_as_double((long long)omn);
            b = __longlong_as_double((long long)omx);
            if (osn) {
                if (!seen) { fmn = a; fmx = b; seen = true; }
                else {
                    if (f64cmp_pg(a, fmn) < 0) fmn = a;
                    if (f64cmp_pg(b, fmx) > 0) fmx = b;
                }
            }
        } else {
            imn = min(imn, omn);
            imx = max(imx, omx);
        }
        xs |= oxs;
    }
    const uint32_t wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) {
        smn[wid] = is_float ? (int64_t)__double_as_longlong(fmn) : imn;
        smx[wid] = is_float ? (int64_t)__double_as_longlong(fmx) : imx;
        sxs[wid] = xs;
        sseen[wid] = seen;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        int64_t bmn = smn[0], bmx = smx[0];
        uint64_t bxs = sxs[0];
        int bseen = sseen[0];
        for (uint32_t w2 = 1; w2 < AGG_BLOCK / WAVE; w2++) {
            if (is_float) {
                if (sseen[w2]) {
                    double a = __longlong_as_double((long long)smn[w2]);
                    double b = __longlong_as_double((long long)smx[w2]);
                    if (!bseen) { bmn = smn[w2]; bmx = smx[w2]; bseen = 1; }
                    else {
                        if (f64cmp_pg(a, __longlong_as_double((long long)bmn)) < 0) bmn = smn[w2];
                        if (f64cmp_pg(b, __longlong_as_double((long long)bmx)) > 0) bmx = smx[w2];
                    }
                }
            } else {
                bmn = min(bmn, smn[w2]);
                bmx = max(bmx, smx[w2]);
            }
            bxs |= sxs[w2];
        }
        /* per-block results merged with CAS loops on the single out cell */
        if (is_float) {
            unsigned long long *amn = (unsigned long long *)&out->mn;
            unsigned long long cur = *amn;
            while (bseen) {
                double c = __longlong_as_double((long long)cur);
                double nv = __longlong_as_double((long long)bmn);
                if (f64cmp_pg(nv, c) >= 0) break;
                unsigned long long prev = atomicCAS(amn, cur, (unsigned long long)bmn);
                if (prev == cur) break;
                cur = prev;
            }
            unsigned long long *amx = (unsigned long long *)&out->mx;
            cur = *amx;
            while (bseen) {
                double c = __longlong_as_double((long long)cur);
                double nv = __longlong_as_double((long long)bmx);
                if (f64cmp_pg(nv, c) <= 0) break;
                unsigned long long prev = atomicCAS(amx, cur, (unsigned long long)bmx);
                if (prev == cur) break;
                cur = prev;
            }
        } else {
            atomicMin((long long *)&out->mn, (long long)bmn);
            atomicMax((long long *)&out->mx, (long long)bmx);
        }
        atomicOr((unsigned long long *)&out->xsig, (unsigned long long)bxs);
    }
}

/* emit the canonical P(L) stream for one width-8 column chunk: lane j
 * writes its value's literal bytes plus the token/offset of the sequence
 * it opens — every byte position is the closed form of format.h */
__global__ __launch_bounds__(AGG_BLOCK) void wemit_p_kernel(
    const uint64_t *__restrict__ v, uint32_t n, uint32_t L,
    uint8_t *__restrict__ out)
{
    const uint32_t j = blockIdx.x * blockDim.x + threadIdx.x;
    if (j >= n) return;
    const uint32_t step = L + 3u;
    const uint8_t mtok = (uint8_t)(8u - L - 4u);
    const uint64_t val = v[j];
    if (j == 0) {
        out[0] = (uint8_t)(((8u + L) << 4) | mtok);
        #pragma unroll
        for (int b = 0; b < 8; b++) out[1 + b] = (uint8_t)(val >> (8 * b));
        /* S0's offset closes after v1's low bytes (lane 1 writes those) */
        out[9 + L] = 8;
        out[10 + L] = 0;
        return;
    }
    if (j == n - 1) {                       /* final: token + full literals */
        const uint32_t fpos = (11u + L) + (n - 3u) * step;
        out[fpos] = (uint8_t)(8u << 4);
        #pragma unroll
        for (int b = 0; b < 8; b++) out[fpos + 1 + b] = (uint8_t)(val >> (8 * b));
        return;
    }
    /* v_j low bytes: j==1 lives in S0's literals at 9; j>=2 at pos(j) with
     * its sequence header [token at pos-1] and trailing offset */
    if (j == 1) {
        for (uint32_t b = 0; b < L; b++) out[9 + b] = (uint8_t)(val >> (8 * b));
        return;
    }
    const uint32_t pos = j * step + (6u - L);
    out[pos - 1] = (uint8_t)((L << 4) | mtok);
    for (uint32_t b = 0; b < L; b++) out[pos + b] = (uint8_t)(val >> (8 * b));
    out[pos + L] = 8;                       /* every S_i carries its offset */
    out[pos + L + 1] = 0;
}

int csgpu_compress_chunk(const void *const *dev_vals, const uint8_t *types,
                         uint32_t n_cols, uint32_t rows, uint64_t row_offset,
                         std::vector<cs_dev_chunk_col> &out)
{
    if (!cstripe_gpu_available()) { cs_set_err("device write path requires a visible MI355X"); return CSTRIPE_ERR_NOGPU; }
    out.assign(n_cols, {});
    hipStream_t st;
    HIP_TRY(hipStreamCreate(&st));
    WStats *d_stats = nullptr;
    uint8_t *d_out = nullptr;
    size_t d_out_cap = 0;
    auto cleanup = [&]() {
        if (d_stats) (void)hipFree(d_stats);
        if (d_out) (void)hipFree(d_out);
        (void)hipStreamDestroy(st);
    };
    #define WTRY(x) do { hipError_t _e = (x); if (_e != hipSuccess) { \
        cs_set_err("%s failed: %s", #x, hipGetErrorString(_e)); cleanup(); return CSTRIPE_ERR; } } while (0)
    WTRY(hipMalloc(&d_stats, sizeof(WStats)));

    for (uint32_t c = 0; c < n_cols; c++) {
        const uint32_t width = csf_type_width(types[c]);
        const bool is_float = types[c] == CSTRIPE_F32 || types[c] == CSTRIPE_F64;
        const uint8_t *dv = (const uint8_t *)dev_vals[c] + row_offset * width;
        cs_dev_chunk_col &oc = out[c];
        oc.has_min_max = rows > 0;

        WStats init;
        if (is_float) {
            const double dnan = 0.0 / 0.0, dninf = -1.0 / 0.0;
            memcpy(&init.mn, &dnan, 8);     /* PG MIN identity: NaN high */
            memcpy(&init.mx, &dninf, 8);
        } else {
            init.mn = INT64_MAX;
            init.mx = INT64_MIN;
        }
        init.xsig = 0;
        WTRY(hipMemcpyAsync(d_stats, &init, sizeof(init), hipMemcpyHostToDevice, st));
        const uint32_t grid = rows ? min((rows + AGG_BLOCK - 1) / AGG_BLOCK, 1024u) : 1;
        if (rows) {
            switch (width) {
                case 1: hipLaunchKernelGGL(wstats_kernel<int8_t>, dim3(grid), dim3(AGG_BLOCK), 0, st, (const int8_t *)dv, rows, is_float, d_stats); break;
                case 2: hipLaunchKernelGGL(wstats_kernel<int16_t>, dim3(grid), dim3(AGG_BLOCK), 0, st, (const int16_t *)dv, rows, is_float, d_stats); break;
                case 4: hipLaunchKernelGGL(wstats_kernel<int32_t>, dim3(grid), dim3(AGG_BLOCK), 0, st, (const int32_t *)dv, rows, is_float, d_stats); break;
                default: hipLaunchKernelGGL(wstats_kernel<int64_t>, dim3(grid), dim3(AGG_BLOCK), 0, st, (const int64_t *)dv, rows, is_float, d_stats); break;
            }
            WTRY(hipGetLastError());
        }
        WStats hs;
        WTRY(hipMemcpyAsync(&hs, d_stats, sizeof(hs), hipMemcpyDeviceToHost, st));
        WTRY(hipStreamSynchronize(st));
        oc.min_i = hs.mn;
        oc.max_i = hs.mx;
        /* TEXT: the host writer records C-collation lex-key min/max
         * (format.h csf_text_lex_key); this stats kernel reduces raw
         * integer order, so device-written TEXT chunks stay conservatively
         * unprunable rather than carry a wrong-order range */
        if (types[c] == CSTRIPE_TEXT) oc.has_min_max = false;

        const uint64_t raw_bytes = (uint64_t)rows * width;
        bool emitted = false;
        if (width == 8 && rows >= 3 && hs.xsig != 0) {
            int hb = 63;
            while (hb > 0 && !((hs.xsig >> hb) & 1)) hb--;
            const uint32_t L = (uint32_t)(hb / 8 + 1);
            if (L <= 4) {
                const uint64_t csz = (11ull + L) + (uint64_t)(rows - 3) * (L + 3) + 9;
                if (csz < raw_bytes) {
                    if (d_out_cap < csz) {
                        if (d_out) WTRY(hipFree(d_out));
                        d_out_cap = (csz + 4095) & ~4095ull;
                        WTRY(hipMalloc(&d_out, d_out_cap));
                    }
                    hipLaunchKernelGGL(wemit_p_kernel,
                                       dim3((rows + AGG_BLOCK - 1) / AGG_BLOCK),
                                       dim3(AGG_BLOCK), 0, st,
                                       (const uint64_t *)dv, rows, L, d_out);
                    WTRY(hipGetLastError());
                    oc.data.resize(csz);
                    WTRY(hipMemcpyAsync(oc.data.data(), d_out, csz, hipMemcpyDeviceToHost, st));
                    WTRY(hipStreamSynchronize(st));
                    oc.mode = (uint8_t)(CSF_SEGMODE_P_BASE | L);
                    oc.canonical = true;
                    emitted = true;
                }
            }
        } else if (width == 8 && rows >= 3 && hs.xsig == 0) {
            /* constant chunk: emit on host from v0 (8 bytes over PCIe) */
            uint64_t v0;
            WTRY(hipMemcpyAsync(&v0, dv, 8, hipMemcpyDeviceToHost, st));
            WTRY(hipStreamSynchronize(st));
            oc.data.resize(64 + rows / 16);
            int csz = lz4e_canon_const((const uint8_t *)&v0, (int)rows,
                                       oc.data.data(), (int)oc.data.size());
            if (csz > 0 && (uint64_t)csz < raw_bytes) {
                oc.data.resize((size_t)csz);
                oc.mode = CSF_SEGMODE_CONST;
                oc.canonical = true;
                emitted = true;
            }
        }
        if (!emitted) {
            /* raw copy-back: the host writer path compresses it */
            oc.data.resize(raw_bytes);
            if (raw_bytes) {
                WTRY(hipMemcpyAsync(oc.data.data(), dv, raw_bytes, hipMemcpyDeviceToHost, st));
                WTRY(hipStreamSynchronize(st));
            }
            oc.canonical = false;
        }
    }
    #undef WTRY
    cleanup();
    return CSTRIPE_OK;
}
