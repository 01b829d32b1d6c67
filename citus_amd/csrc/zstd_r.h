/*
 * zstd_r.h — RESTRICTED zstd codec for device-decodable column chunks.
 *
 * The writer emits STANDARD zstd frames (RFC 8878) restricted to the
 * features a GPU lane can decode without per-frame table construction:
 *   - single-segment frames, no checksum, no dictionary;
 *   - one compressed block (or a raw block when compression loses);
 *   - RAW literals (no Huffman);
 *   - sequences entropy-coded with the PREDEFINED FSE distributions
 *     (Predefined_Mode for LL/OF/ML — the decode tables are fixed by the
 *     spec, so both sides build them once from the distributions below);
 *   - no repeat-offset codes (offset_value = offset + 3 always).
 *
 * Every emitted frame is decodable by the system libzstd — pinned by
 * tests/test_format.py::test_zstd_restricted_system_decodable — so the
 * on-disk artifact remains plain zstd (the reference's DecompressBuffer,
 * columnar_compression.c:207, consumes it unchanged). Frames from OTHER
 * producers (libzstd level-3 output, foreign files) are host-decoded; only
 * segments tagged CSF_SEGMODE_ZR (writer-emitted restricted frames) take
 * the device path.
 *
 * Everything below is written from the published RFC 8878 format
 * description (frame/block headers, FSE table spread + state machine,
 * sequence code tables), not from zstd sources.
 */
#ifndef CSTRIPE_ZSTD_R_H
#define CSTRIPE_ZSTD_R_H

#include <stdint.h>
#include <string.h>

#ifndef ZR_HOSTDEV
#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define ZR_HOSTDEV __host__ __device__
#else
#define ZR_HOSTDEV
#endif
#endif

/* ---------------- predefined distributions (RFC 8878) ---------------- */

#define ZR_LL_ACCLOG 6
#define ZR_ML_ACCLOG 6
#define ZR_OF_ACCLOG 5
#define ZR_LL_SYMS 36
#define ZR_ML_SYMS 53
#define ZR_OF_SYMS 29

static const int16_t ZR_LL_NORM[ZR_LL_SYMS] = {
    4, 3, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 1, 1, 1,
    2, 2, 2, 2, 2, 2, 2, 2, 2, 3, 2, 1, 1, 1, 1, 1,
    -1, -1, -1, -1};
static const int16_t ZR_ML_NORM[ZR_ML_SYMS] = {
    1, 4, 3, 2, 2, 2, 2, 2, 2, 1, 1, 1, 1, 1, 1, 1,
    1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
    1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, -1, -1,
    -1, -1, -1, -1, -1};
static const int16_t ZR_OF_NORM[ZR_OF_SYMS] = {
    1, 1, 1, 1, 1, 1, 2, 2, 2, 1, 1, 1, 1, 1, 1, 1,
    1, 1, 1, 1, 1, 1, 1, 1, -1, -1, -1, -1, -1};

/* LL code -> (baseline, nbits); codes 0-15 are direct */
static const uint32_t ZR_LL_BASE[ZR_LL_SYMS] = {
    0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15,
    16, 18, 20, 22, 24, 28, 32, 40, 48, 64, 128, 256, 512, 1024, 2048, 4096,
    8192, 16384, 32768, 65536};
static const uint8_t ZR_LL_BITS[ZR_LL_SYMS] = {
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    1, 1, 1, 1, 2, 2, 3, 3, 4, 6, 7, 8, 9, 10, 11, 12,
    13, 14, 15, 16};

/* ML code -> (baseline, nbits); codes 0-31 are ml = code + 3 */
static const uint32_t ZR_ML_BASE[ZR_ML_SYMS] = {
    3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16, 17, 18,
    19, 20, 21, 22, 23, 24, 25, 26, 27, 28, 29, 30, 31, 32, 33, 34,
    35, 37, 39, 41, 43, 47, 51, 59, 67, 83, 99, 131, 259, 515, 1027, 2051,
    4099, 8195, 16387, 32771, 65539};
static const uint8_t ZR_ML_BITS[ZR_ML_SYMS] = {
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    1, 1, 1, 1, 2, 2, 3, 3, 4, 4, 5, 7, 8, 9, 10, 11,
    12, 13, 14, 15, 16};

/* ---------------- FSE decode table (spec construction) ---------------- */

typedef struct {
    uint8_t symbol;
    uint8_t nbits;
    uint16_t base;          /* newState baseline */
} zr_dcell;

typedef struct {
    zr_dcell ll[1 << ZR_LL_ACCLOG];
    zr_dcell ml[1 << ZR_ML_ACCLOG];
    zr_dcell of[1 << ZR_OF_ACCLOG];
} zr_dtables;

static inline int zr_highbit(uint32_t v)
{
    int r = 0;
    while (v > 1) { v >>= 1; r++; }
    return r;
}

static inline void zr_build_dtable(zr_dcell *t, const int16_t *norm,
                                   int nsyms, int acclog)
{
    const int size = 1 << acclog;
    int high = size - 1;
    uint16_t cnt[64];
    for (int s = 0; s < nsyms; s++) {
        if (norm[s] == -1) {
            t[high--].symbol = (uint8_t)s;
            cnt[s] = 1;
        } else {
            cnt[s] = (uint16_t)norm[s];
        }
    }
    const int step = (size >> 1) + (size >> 3) + 3;
    int pos = 0;
    for (int s = 0; s < nsyms; s++) {
        if (norm[s] <= 0) continue;
        for (int i = 0; i < norm[s]; i++) {
            t[pos].symbol = (uint8_t)s;
            do { pos = (pos + step) & (size - 1); } while (pos > high);
        }
    }
    uint16_t next[64];
    for (int s = 0; s < nsyms; s++) next[s] = cnt[s];
    for (int u = 0; u < size; u++) {
        const int s = t[u].symbol;
        const uint16_t ns = next[s]++;
        const int nb = acclog - zr_highbit(ns);
        t[u].nbits = (uint8_t)nb;
        t[u].base = (uint16_t)((ns << nb) - size);
    }
}

static inline void zr_build_dtables(zr_dtables *dt)
{
    zr_build_dtable(dt->ll, ZR_LL_NORM, ZR_LL_SYMS, ZR_LL_ACCLOG);
    zr_build_dtable(dt->ml, ZR_ML_NORM, ZR_ML_SYMS, ZR_ML_ACCLOG);
    zr_build_dtable(dt->of, ZR_OF_NORM, ZR_OF_SYMS, ZR_OF_ACCLOG);
}

/* ---------------- FSE encode table (encoder side) ---------------- */

typedef struct {
    int32_t delta_nbbits[64];     /* (maxBits << 16) - minStatePlus */
    int32_t delta_find[64];
    uint16_t state_table[64];     /* tableSize entries: next state + size */
    int acclog;
} zr_ctable;

static inline void zr_build_ctable(zr_ctable *c, const int16_t *norm,
                                   int nsyms, int acclog)
{
    const int size = 1 << acclog;
    c->acclog = acclog;
    /* spread identical to the decoder's */
    uint8_t spread[64];
    {
        int high = size - 1;
        for (int s = 0; s < nsyms; s++)
            if (norm[s] == -1) spread[high--] = (uint8_t)s;
        const int step = (size >> 1) + (size >> 3) + 3;
        int pos = 0;
        for (int s = 0; s < nsyms; s++) {
            if (norm[s] <= 0) continue;
            for (int i = 0; i < norm[s]; i++) {
                spread[pos] = (uint8_t)s;
                do { pos = (pos + step) & (size - 1); } while (pos > high);
            }
        }
    }
    int cumul[65];
    cumul[0] = 0;
    for (int s = 0; s < nsyms; s++)
        cumul[s + 1] = cumul[s] + (norm[s] == -1 ? 1 : norm[s]);
    int fill[64];
    for (int s = 0; s < nsyms; s++) fill[s] = cumul[s];
    for (int u = 0; u < size; u++)
        c->state_table[fill[spread[u]]++] = (uint16_t)(size + u);
    int total = 0;
    for (int s = 0; s < nsyms; s++) {
        if (norm[s] == 0) {
            c->delta_nbbits[s] = ((acclog + 1) << 16) - (1 << acclog);
            c->delta_find[s] = 0;
        } else if (norm[s] == -1 || norm[s] == 1) {
            c->delta_nbbits[s] = (acclog << 16) - (1 << acclog);
            c->delta_find[s] = total - 1;
            total += 1;
        } else {
            const int max_bits = acclog - zr_highbit((uint32_t)(norm[s] - 1));
            const int min_state_plus = norm[s] << max_bits;
            c->delta_nbbits[s] = (max_bits << 16) - min_state_plus;
            c->delta_find[s] = total - norm[s];
            total += norm[s];
        }
    }
}

typedef struct {
    zr_ctable ll, ml, of;
} zr_ctables;

static inline void zr_build_ctables(zr_ctables *ct)
{
    zr_build_ctable(&ct->ll, ZR_LL_NORM, ZR_LL_SYMS, ZR_LL_ACCLOG);
    zr_build_ctable(&ct->ml, ZR_ML_NORM, ZR_ML_SYMS, ZR_ML_ACCLOG);
    zr_build_ctable(&ct->of, ZR_OF_NORM, ZR_OF_SYMS, ZR_OF_ACCLOG);
}

/* ---------------- sequence code helpers ---------------- */

static inline uint8_t zr_ll_code(uint32_t ll)
{
    if (ll < 16) return (uint8_t)ll;
    for (int c = ZR_LL_SYMS - 1; c >= 16; c--)
        if (ll >= ZR_LL_BASE[c]) return (uint8_t)c;
    return 16;
}

static inline uint8_t zr_ml_code(uint32_t ml)   /* ml >= 3 */
{
    if (ml <= 34) return (uint8_t)(ml - 3);
    for (int c = ZR_ML_SYMS - 1; c >= 32; c--)
        if (ml >= ZR_ML_BASE[c]) return (uint8_t)c;
    return 32;
}

static inline uint8_t zr_of_code(uint32_t offset_value)
{
    return (uint8_t)zr_highbit(offset_value);
}

/* ---------------- bit writer (LE, final 1-bit end marker) ------------- */

typedef struct {
    uint8_t *out;
    int cap;
    int pos;
    uint64_t acc;
    int nbits;
} zr_bw;

static inline void zr_bw_init(zr_bw *b, uint8_t *out, int cap)
{
    b->out = out; b->cap = cap; b->pos = 0; b->acc = 0; b->nbits = 0;
}

static inline int zr_bw_add(zr_bw *b, uint32_t val, int nbits)
{
    b->acc |= ((uint64_t)(val & ((1u << nbits) - 1u))) << b->nbits;
    b->nbits += nbits;
    while (b->nbits >= 8) {
        if (b->pos >= b->cap) return -1;
        b->out[b->pos++] = (uint8_t)b->acc;
        b->acc >>= 8;
        b->nbits -= 8;
    }
    return 0;
}

static inline int zr_bw_close(zr_bw *b)   /* end marker + pad */
{
    if (zr_bw_add(b, 1, 1) < 0) return -1;
    if (b->nbits > 0) {
        if (b->pos >= b->cap) return -1;
        b->out[b->pos++] = (uint8_t)b->acc;
        b->acc = 0; b->nbits = 0;
    }
    return b->pos;
}

/* ---------------- FSE encoder state ---------------- */

typedef struct {
    uint32_t value;
} zr_cstate;

static inline void zr_cstate_init(zr_cstate *st, const zr_ctable *ct, uint8_t sym)
{
    const uint32_t nbits_out = (uint32_t)((ct->delta_nbbits[sym] + (1 << 15)) >> 16);
    const uint32_t base = ((nbits_out << 16) - (uint32_t)ct->delta_nbbits[sym]);
    st->value = ct->state_table[(base >> nbits_out) + ct->delta_find[sym]];
}

static inline int zr_cstate_encode(zr_cstate *st, const zr_ctable *ct,
                                   uint8_t sym, zr_bw *bw)
{
    const uint32_t nbits_out =
        (uint32_t)(st->value + (uint32_t)ct->delta_nbbits[sym]) >> 16;
    if (zr_bw_add(bw, st->value, (int)nbits_out) < 0) return -1;
    st->value = ct->state_table[(st->value >> nbits_out) + ct->delta_find[sym]];
    return 0;
}

static inline int zr_cstate_flush(zr_cstate *st, const zr_ctable *ct, zr_bw *bw)
{
    return zr_bw_add(bw, st->value, ct->acclog);
}

/* ---------------- greedy matcher (zstd min-match 3) ---------------- */

#define ZR_MAX_SEQS 4096
#define ZR_HASH_LOG 12

typedef struct {
    uint32_t ll;        /* literal run before the match */
    uint32_t off;       /* match offset (actual) */
    uint32_t ml;        /* match length >= 3 */
} zr_seq;

/* ---------------- frame encoder ---------------- */

/* returns frame size, or 0 when dst is too small / input empty */
static inline int zr_compress(const uint8_t *src, int slen,
                              uint8_t *dst, int dcap, const zr_ctables *ct)
{
    /* zstd blocks carry at most 128 KB of regenerated content; this
     * encoder emits ONE block per frame, so segments are capped below
     * that (the writer splits chunks accordingly) */
    if (slen <= 0 || slen > (110 << 10)) return 0;
    int op = 0;
    /* frame header: magic + FHD (single segment) + content size */
    if (op + 9 > dcap) return 0;
    dst[op++] = 0x28; dst[op++] = 0xB5; dst[op++] = 0x2F; dst[op++] = 0xFD;
    if (slen <= 255) {
        dst[op++] = 0x20;                 /* FHD: single_segment, FCS 1 byte */
        dst[op++] = (uint8_t)slen;
    } else if (slen <= 65535 + 256) {
        dst[op++] = 0x60;                 /* FHD: single_segment, FCS 2 bytes
                                           * (stores size - 256; caps 65791) */
        const uint32_t f = (uint32_t)slen - 256;
        dst[op++] = (uint8_t)f;
        dst[op++] = (uint8_t)(f >> 8);
    } else {
        dst[op++] = 0xA0;                 /* FHD: single_segment, FCS 4 bytes */
        const uint32_t f = (uint32_t)slen;
        dst[op++] = (uint8_t)f;
        dst[op++] = (uint8_t)(f >> 8);
        dst[op++] = (uint8_t)(f >> 16);
        dst[op++] = (uint8_t)(f >> 24);
    }
    const int bh_pos = op;                /* 3-byte block header backpatched */
    op += 3;

    /* greedy parse */
    static_assert(ZR_MAX_SEQS >= 2, "");
    zr_seq seqs[ZR_MAX_SEQS];
    int nseq = 0;
    uint32_t lit_total = 0, trailing = 0;
    {
        int32_t table[1 << ZR_HASH_LOG];
        for (int i = 0; i < (1 << ZR_HASH_LOG); i++) table[i] = -1;
        int anchor = 0, pos = 0;
        const int limit = slen - 5;       /* keep a literal tail */
        while (pos < limit && nseq < ZR_MAX_SEQS) {
            uint32_t v;
            memcpy(&v, src + pos, 4);
            const uint32_t h = (v * 2654435761u) >> (32 - ZR_HASH_LOG);
            const int32_t cand = table[h];
            table[h] = pos;
            if (cand >= 0 && pos - cand <= 131072) {
                uint32_t cv;
                memcpy(&cv, src + cand, 4);
                if (cv == v) {
                    int ml = 4;
                    while (pos + ml < slen - 1 && src[cand + ml] == src[pos + ml])
                        ml++;
                    if (ml > 131070) ml = 131070;   /* ML code ceiling */
                    if (pos - anchor > 131000)
                        return 0;   /* LL code ceiling: fall back to a raw
                                     * frame rather than an unencodable run */
                    seqs[nseq].ll = (uint32_t)(pos - anchor);
                    seqs[nseq].off = (uint32_t)(pos - cand);
                    seqs[nseq].ml = (uint32_t)ml;
                    nseq++;
                    lit_total += (uint32_t)(pos - anchor);
                    pos += ml;
                    anchor = pos;
                    continue;
                }
            }
            pos++;
        }
        trailing = (uint32_t)(slen - anchor);
        lit_total += trailing;
    }


    if (nseq == 0) {
        /* nothing found: raw block */
        const uint32_t bh = 1u | (0u << 1) | ((uint32_t)slen << 3);
        dst[bh_pos] = (uint8_t)bh;
        dst[bh_pos + 1] = (uint8_t)(bh >> 8);
        dst[bh_pos + 2] = (uint8_t)(bh >> 16);
        if (op + slen > dcap) return 0;
        memcpy(dst + op, src, (size_t)slen);
        return op + slen;
    }

    /* literals section: RAW literals */
    {
        int need = lit_total <= 31 ? 1 : (lit_total <= 4095 ? 2 : 3);
        if (op + need + (int)lit_total > dcap) return 0;
        if (lit_total <= 31) {
            dst[op++] = (uint8_t)(lit_total << 3);
        } else if (lit_total <= 4095) {
            dst[op++] = (uint8_t)(0x04 | ((lit_total & 0x0F) << 4));
            dst[op++] = (uint8_t)(lit_total >> 4);
        } else {
            dst[op++] = (uint8_t)(0x0C | ((lit_total & 0x0F) << 4));
            dst[op++] = (uint8_t)(lit_total >> 4);
            dst[op++] = (uint8_t)(lit_total >> 12);
        }
        /* literal bytes: per-sequence runs then the trailing run */
        int ip = 0;
        for (int n = 0; n < nseq; n++) {
            memcpy(dst + op, src + ip, seqs[n].ll);
            op += (int)seqs[n].ll;
            ip += (int)(seqs[n].ll + seqs[n].ml);
        }
        memcpy(dst + op, src + ip, trailing);
        op += (int)trailing;
    }

    /* sequences section */
    {
        if (op + 4 > dcap) return 0;
        if (nseq < 128) {
            dst[op++] = (uint8_t)nseq;
        } else {
            dst[op++] = (uint8_t)((nseq >> 8) + 128);
            dst[op++] = (uint8_t)nseq;
        }
        dst[op++] = 0x00;                 /* all predefined modes */
        zr_bw bw;
        zr_bw_init(&bw, dst + op, dcap - op);
        const int last = nseq - 1;
        uint8_t llc[ZR_MAX_SEQS], mlc[ZR_MAX_SEQS], ofc[ZR_MAX_SEQS];
        for (int n = 0; n < nseq; n++) {
            llc[n] = zr_ll_code(seqs[n].ll);
            mlc[n] = zr_ml_code(seqs[n].ml);
            ofc[n] = zr_of_code(seqs[n].off + 3);
        }
        zr_cstate sml, sof, sll;
        zr_cstate_init(&sml, &ct->ml, mlc[last]);
        zr_cstate_init(&sof, &ct->of, ofc[last]);
        zr_cstate_init(&sll, &ct->ll, llc[last]);
        /* last sequence's extra bits */
        if (zr_bw_add(&bw, seqs[last].ll - ZR_LL_BASE[llc[last]], ZR_LL_BITS[llc[last]]) < 0) return 0;
        if (zr_bw_add(&bw, seqs[last].ml - ZR_ML_BASE[mlc[last]], ZR_ML_BITS[mlc[last]]) < 0) return 0;
        {
            const uint32_t ov = seqs[last].off + 3;
            if (zr_bw_add(&bw, ov - (1u << ofc[last]), ofc[last]) < 0) return 0;
        }
        for (int n = last - 1; n >= 0; n--) {
            if (zr_cstate_encode(&sof, &ct->of, ofc[n], &bw) < 0) return 0;
            if (zr_cstate_encode(&sml, &ct->ml, mlc[n], &bw) < 0) return 0;
            if (zr_cstate_encode(&sll, &ct->ll, llc[n], &bw) < 0) return 0;
            if (zr_bw_add(&bw, seqs[n].ll - ZR_LL_BASE[llc[n]], ZR_LL_BITS[llc[n]]) < 0) return 0;
            if (zr_bw_add(&bw, seqs[n].ml - ZR_ML_BASE[mlc[n]], ZR_ML_BITS[mlc[n]]) < 0) return 0;
            {
                const uint32_t ov = seqs[n].off + 3;
                if (zr_bw_add(&bw, ov - (1u << ofc[n]), ofc[n]) < 0) return 0;
            }
        }
        if (zr_cstate_flush(&sml, &ct->ml, &bw) < 0) return 0;
        if (zr_cstate_flush(&sof, &ct->of, &bw) < 0) return 0;
        if (zr_cstate_flush(&sll, &ct->ll, &bw) < 0) return 0;
        const int bs = zr_bw_close(&bw);
        if (bs < 0) return 0;
        op += bs;
    }

    const uint32_t bsize = (uint32_t)(op - bh_pos - 3);
    if (bsize > 0x1FFFFF) return 0;
    const uint32_t bh = 1u | (2u << 1) | (bsize << 3);
    dst[bh_pos] = (uint8_t)bh;
    dst[bh_pos + 1] = (uint8_t)(bh >> 8);
    dst[bh_pos + 2] = (uint8_t)(bh >> 16);
    return op < slen ? op : 0;            /* only keep it if it shrank */
}

/* ---------------- shared host/device decoder (restricted frames) -------- */

/* backward bit reader over [src, src+len) with a cached 64-bit window —
 * the cursor only moves DOWN, so one unaligned 8-byte load serves several
 * reads (the naive per-read byte loop was the device decoder's bottleneck:
 * ~5 dependent global loads per bit-read) */
typedef struct {
    const uint8_t *src;
    int32_t len;
    int32_t bitpos;       /* bits remaining below the cursor */
    int32_t wbase;        /* byte index of the window's LSB (-1 = empty) */
    uint64_t w;
} zr_br;

ZR_HOSTDEV static inline int zr_br_init(zr_br *b, const uint8_t *src, int len)
{
    b->src = src;
    b->len = len;
    b->wbase = -1;
    b->w = 0;
    int last = len - 1;
    while (last >= 0 && src[last] == 0) last--;
    if (last < 0) return -1;
    int hb = 7;
    while (!(src[last] & (1 << hb))) hb--;
    b->bitpos = last * 8 + hb;            /* marker bit excluded */
    return 0;
}

ZR_HOSTDEV static inline uint32_t zr_br_read(zr_br *b, int nbits)
{
    if (nbits == 0) return 0;
    b->bitpos -= nbits;
    const int32_t bp = b->bitpos < 0 ? 0 : b->bitpos;
    const int32_t byte = bp >> 3;
    if (b->wbase < 0 || byte < b->wbase ||
        (bp - b->wbase * 8) + nbits > 64) {
        int32_t base = byte - 3;          /* room for a few more reads below */
        if (base > b->len - 8) base = b->len - 8;
        if (base < 0) base = 0;
        if (b->len >= 8) {
            uint64_t w;
            __builtin_memcpy(&w, b->src + base, 8);
            b->w = w;
        } else {
            uint64_t w = 0;
            for (int i = 0; i < b->len; i++)
                w |= (uint64_t)b->src[i] << (8 * i);
            b->w = w;
            base = 0;
        }
        b->wbase = base;
    }
    const int32_t sh = bp - b->wbase * 8;
    return (uint32_t)(b->w >> sh) & ((1u << nbits) - 1u);
}

/* decode ONE restricted frame; returns decompressed size or -1 */
ZR_HOSTDEV static inline int zr_decode_frame(const uint8_t *src, int slen,
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
    if (!(fhd == 0x20 || fhd == 0x60 || fhd == 0xA0)) return -1;  /* restricted */
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
    /* block header */
    const uint32_t bh = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8) |
                        ((uint32_t)src[ip + 2] << 16);
    ip += 3;
    const int btype = (int)((bh >> 1) & 3);
    const int bsize = (int)(bh >> 3);
    if (ip + bsize > slen) return -1;
    if (btype == 0) {                     /* raw block */
        if (bsize != content) return -1;
        for (int i = 0; i < content; i++) dst[i] = src[ip + i];
        return content;
    }
    if (btype != 2) return -1;
    const int bend = ip + bsize;

    /* literals section: RAW only */
    const uint8_t lh = src[ip];
    if ((lh & 3) != 0) return -1;
    int lit_size;
    if (!(lh & 0x04)) { lit_size = lh >> 3; ip += 1; }
    else if (!(lh & 0x08)) { lit_size = (lh >> 4) | ((int)src[ip + 1] << 4); ip += 2; }
    else { lit_size = (lh >> 4) | ((int)src[ip + 1] << 4) | ((int)src[ip + 2] << 12); ip += 3; }
    const uint8_t *lits = src + ip;
    ip += lit_size;
    if (ip > bend) return -1;

    /* sequences */
    int nseq = src[ip++];
    if (nseq >= 128) {
        if (nseq == 255) { nseq = src[ip] + (src[ip + 1] << 8) + 0x7F00; ip += 2; }
        else { nseq = ((nseq - 128) << 8) + src[ip]; ip += 1; }
    }
    int op = 0, lp = 0;
    if (nseq == 0) {
        if (lit_size != content) return -1;
        for (int i = 0; i < content; i++) dst[i] = lits[i];
        return content;
    }
    if (src[ip++] != 0x00) return -1;     /* predefined modes only */

    zr_br br;
    if (zr_br_init(&br, src + ip, bend - ip) < 0) return -1;
    uint32_t sll = zr_br_read(&br, ZR_LL_ACCLOG);
    uint32_t sof = zr_br_read(&br, ZR_OF_ACCLOG);
    uint32_t sml = zr_br_read(&br, ZR_ML_ACCLOG);

    for (int n = 0; n < nseq; n++) {
        const zr_dcell cll = llt[sll], cml = mlt[sml], cof = oft[sof];
        const uint32_t of_val = (1u << cof.symbol) + zr_br_read(&br, cof.symbol);
        uint32_t ml = ZR_ML_BASE[cml.symbol] + zr_br_read(&br, ZR_ML_BITS[cml.symbol]);
        uint32_t ll = ZR_LL_BASE[cll.symbol] + zr_br_read(&br, ZR_LL_BITS[cll.symbol]);
        if (n + 1 < nseq) {
            sll = cll.base + zr_br_read(&br, cll.nbits);
            sml = cml.base + zr_br_read(&br, cml.nbits);
            sof = cof.base + zr_br_read(&br, cof.nbits);
        }
        if (of_val <= 3) return -1;       /* repeat offsets not emitted */
        const uint32_t off = of_val - 3;
        if (lp + (int)ll > lit_size || op + (int)(ll + ml) > content) return -1;
        {   /* word-wise literal copy (unaligned 8B ok on gfx950 and host) */
            uint32_t i = 0;
            for (; i + 8 <= ll; i += 8) {
                uint64_t v;
                __builtin_memcpy(&v, lits + lp + (int)i, 8);
                __builtin_memcpy(dst + op + (int)i, &v, 8);
            }
            for (; i < ll; i++) dst[op + (int)i] = lits[lp + (int)i];
        }
        op += (int)ll; lp += (int)ll;
        if (off > (uint32_t)op) return -1;
        {
            uint32_t i = 0;
            if (off >= 8) {               /* no overlap within a word */
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
    {
        int i = 0;
        for (; i + 8 <= rem; i += 8) {
            uint64_t v;
            __builtin_memcpy(&v, lits + lp + i, 8);
            __builtin_memcpy(dst + op + i, &v, 8);
        }
        for (; i < rem; i++) dst[op + i] = lits[lp + i];
    }
    return content;
}

/* ---------------- canonical frame emitters (closed-form GPU access) ----
 * The zstd analogue of lz4_enc.h's canonical parses: a VALID restricted
 * frame (standard RFC 8878, raw literals + predefined-FSE sequences —
 * ZSTD_decompress-decodable, pinned in tests) whose literal section lays
 * the values out at CLOSED-FORM positions, so the GPU reads values straight
 * from the compressed stream and never walks the FSE chain (the per-lane
 * serial decode is the instruction wall, round-1 VERDICT #1 for LZ4).
 * Fixed header shape: magic(4) + FHD 0xA0(1) + FCS(4) + block header(3) +
 * 3-byte literals header = 15 bytes to the first literal byte.
 *   P(L):  literals = v0 (8 B) + low-L bytes of v1..v_{n-1};
 *          sequences = S0 (ll=8+L, ml=8-L, off=8) then n-2 x (ll=L,
 *          ml=8-L, off=8); value 0's bytes at 15, value j>=1's low-L bytes
 *          at 23 + (j-1)*L (csf_canon_zrp_pos)
 *   CONST: literals = v0 (8 B); one sequence (ll=8, ml=8(n-1), off=8)
 * Requires 1 <= L <= 4, n >= 3, 8n < 127 KB (zstd block content cap). */
#define ZR_CANON_LIT0 15

static inline int zr_canon_hdr(uint8_t *dst, int dcap, uint32_t content,
                               uint32_t lit_size)
{
    if (dcap < ZR_CANON_LIT0) return -1;
    int op = 0;
    dst[op++] = 0x28; dst[op++] = 0xB5; dst[op++] = 0x2F; dst[op++] = 0xFD;
    dst[op++] = 0xA0;                      /* single_segment, FCS 4 bytes */
    dst[op++] = (uint8_t)content; dst[op++] = (uint8_t)(content >> 8);
    dst[op++] = (uint8_t)(content >> 16); dst[op++] = (uint8_t)(content >> 24);
    op += 3;                               /* block header backpatched */
    dst[op++] = (uint8_t)(0x0C | ((lit_size & 0x0F) << 4));   /* RAW, 3-byte */
    dst[op++] = (uint8_t)(lit_size >> 4);
    dst[op++] = (uint8_t)(lit_size >> 12);
    return op;                             /* == ZR_CANON_LIT0 */
}

static inline int zr_canon_close(uint8_t *dst, int op)   /* backpatch bh */
{
    const uint32_t bsize = (uint32_t)(op - 9 - 3);
    if (bsize > 0x1FFFFF) return 0;
    const uint32_t bh = 1u | (2u << 1) | (bsize << 3);     /* last, compressed */
    dst[9] = (uint8_t)bh; dst[10] = (uint8_t)(bh >> 8); dst[11] = (uint8_t)(bh >> 16);
    return op;
}

/* canonical P(L): n width-8 values sharing their high (8-L) bytes */
static inline int zr_canon_p(const uint8_t *src, int n, int L,
                             uint8_t *dst, int dcap, const zr_ctables *ct)
{
    if (n < 3 || L < 1 || L > 4 || (int64_t)n * 8 >= (127 << 10)) return 0;
    const uint32_t content = (uint32_t)n * 8;
    const uint32_t lit_size = 8 + (uint32_t)(n - 1) * (uint32_t)L;
    int op = zr_canon_hdr(dst, dcap, content, lit_size);
    if (op < 0 || op + (int)lit_size + 4 > dcap) return 0;
    memcpy(dst + op, src, 8);              /* v0 full */
    op += 8;
    for (int j = 1; j < n; j++) {
        memcpy(dst + op, src + (size_t)j * 8, (size_t)L);
        op += L;
    }
    const int nseq = n - 1;
    if (nseq < 128) dst[op++] = (uint8_t)nseq;
    else { dst[op++] = (uint8_t)((nseq >> 8) + 128); dst[op++] = (uint8_t)nseq; }
    dst[op++] = 0x00;                      /* predefined modes */
    const uint8_t llc_first = (uint8_t)(8 + L), llc_rest = (uint8_t)L;
    const uint8_t mlc = (uint8_t)(8 - L - 3);
    const uint8_t ofc = 3;                 /* off 8 -> value 11: 3 extra bits */
    zr_bw bw;
    zr_bw_init(&bw, dst + op, dcap - op);
    zr_cstate sml, sof, sll;
    zr_cstate_init(&sml, &ct->ml, mlc);
    zr_cstate_init(&sof, &ct->of, ofc);
    zr_cstate_init(&sll, &ct->ll, llc_rest);            /* last seq (n>=3) */
    if (zr_bw_add(&bw, 3, 3) < 0) return 0;             /* last seq OF extra
                                                         * (LL/ML codes here
                                                         * carry 0 bits) */
    for (int i = nseq - 2; i >= 0; i--) {
        if (zr_cstate_encode(&sof, &ct->of, ofc, &bw) < 0) return 0;
        if (zr_cstate_encode(&sml, &ct->ml, mlc, &bw) < 0) return 0;
        if (zr_cstate_encode(&sll, &ct->ll, i == 0 ? llc_first : llc_rest, &bw) < 0) return 0;
        if (zr_bw_add(&bw, 3, 3) < 0) return 0;
    }
    if (zr_cstate_flush(&sml, &ct->ml, &bw) < 0) return 0;
    if (zr_cstate_flush(&sof, &ct->of, &bw) < 0) return 0;
    if (zr_cstate_flush(&sll, &ct->ll, &bw) < 0) return 0;
    const int bs = zr_bw_close(&bw);
    if (bs < 0) return 0;
    return zr_canon_close(dst, op + bs);
}

/* canonical constant: n equal width-8 values */
static inline int zr_canon_const(const uint8_t *src, int n,
                                 uint8_t *dst, int dcap, const zr_ctables *ct)
{
    if (n < 3 || (int64_t)n * 8 >= (127 << 10)) return 0;
    const uint32_t content = (uint32_t)n * 8;
    int op = zr_canon_hdr(dst, dcap, content, 8);
    if (op < 0 || op + 40 > dcap) return 0;
    memcpy(dst + op, src, 8);              /* v0 */
    op += 8;
    dst[op++] = 1;                         /* nseq */
    dst[op++] = 0x00;
    const uint32_t ml = (uint32_t)(n - 1) * 8;
    const uint8_t llc = 8, mlc = zr_ml_code(ml), ofc = 3;
    zr_bw bw;
    zr_bw_init(&bw, dst + op, dcap - op);
    zr_cstate sml, sof, sll;
    zr_cstate_init(&sml, &ct->ml, mlc);
    zr_cstate_init(&sof, &ct->of, ofc);
    zr_cstate_init(&sll, &ct->ll, llc);
    if (zr_bw_add(&bw, ml - ZR_ML_BASE[mlc], ZR_ML_BITS[mlc]) < 0) return 0;
    if (zr_bw_add(&bw, 3, 3) < 0) return 0;
    if (zr_cstate_flush(&sml, &ct->ml, &bw) < 0) return 0;
    if (zr_cstate_flush(&sof, &ct->of, &bw) < 0) return 0;
    if (zr_cstate_flush(&sll, &ct->ll, &bw) < 0) return 0;
    const int bs = zr_bw_close(&bw);
    if (bs < 0) return 0;
    return zr_canon_close(dst, op + bs);
}

/* ---------------- width-4 canonical frames (char(1) flag columns) -------
 * A width-4 slot stream (short-varlena char(1): [0x05 ch 00 00]) where ALL
 * slots share three of their four bytes and differ only in byte k. zstd's
 * 3-byte minimum match makes a closed-form parse possible that LZ4 (min
 * match 4) cannot express: per row, ONE literal byte (the varying byte)
 * and one ml=3 off=4 match reproducing the 3 shared bytes that separate
 * consecutive varying bytes in the stream.
 *   literals = [slot0 (4 B) + pre1 (k B) + b_1] + [b_2 .. b_{n-2}] +
 *              [b_{n-1} + post_{n-1} (3-k B)]          (lit_size = n + 6)
 *   sequences = S0 (ll=5+k, ml=3, off=4) then n-3 x (ll=1, ml=3, off=4)
 *               — nseq = n-2 — then 4-k trailing literals
 * Varying byte of row j at lit position (j==0 ? k : j+3+k); the slot is
 * hval | (b_j << 8k) with hval = slot0 & ~(0xFF << 8k). Also the all-equal
 * width-4 case as one big ml=4(n-1) off=4 match (ZR4 CONST). */

static inline int zr_canon_b4(const uint8_t *src, int n, int k,
                              uint8_t *dst, int dcap, const zr_ctables *ct)
{
    if (n < 3 || k < 0 || k > 3 || (int64_t)n * 4 >= (127 << 10)) return 0;
    const uint32_t content = (uint32_t)n * 4;
    const uint32_t lit_size = (uint32_t)n + 6;
    int op = zr_canon_hdr(dst, dcap, content, lit_size);
    if (op < 0 || op + (int)lit_size + 4 > dcap) return 0;
    memcpy(dst + op, src, 4 + (size_t)k);    /* slot0 + pre of slot1 */
    op += 4 + k;
    dst[op++] = src[4 + (size_t)k];          /* b_1 */
    for (int j = 2; j <= n - 2; j++)
        dst[op++] = src[(size_t)j * 4 + k];
    dst[op++] = src[(size_t)(n - 1) * 4 + k];
    memcpy(dst + op, src + (size_t)(n - 1) * 4 + k + 1, (size_t)(3 - k));
    op += 3 - k;
    const int nseq = n - 2;
    if (nseq < 128) dst[op++] = (uint8_t)nseq;
    else { dst[op++] = (uint8_t)((nseq >> 8) + 128); dst[op++] = (uint8_t)nseq; }
    dst[op++] = 0x00;
    const uint8_t llc_first = (uint8_t)(5 + k), llc_rest = 1;
    const uint8_t mlc = 0;                   /* ml 3 */
    const uint8_t ofc = 2;                   /* off 4 -> value 7: 2 extra bits */
    zr_bw bw;
    zr_bw_init(&bw, dst + op, dcap - op);
    zr_cstate sml, sof, sll;
    zr_cstate_init(&sml, &ct->ml, mlc);
    zr_cstate_init(&sof, &ct->of, ofc);
    zr_cstate_init(&sll, &ct->ll, nseq == 1 ? llc_first : llc_rest);
    if (zr_bw_add(&bw, 3, 2) < 0) return 0;  /* last seq OF extra (7-4) */
    for (int i = nseq - 2; i >= 0; i--) {
        if (zr_cstate_encode(&sof, &ct->of, ofc, &bw) < 0) return 0;
        if (zr_cstate_encode(&sml, &ct->ml, mlc, &bw) < 0) return 0;
        if (zr_cstate_encode(&sll, &ct->ll, i == 0 ? llc_first : llc_rest, &bw) < 0) return 0;
        if (zr_bw_add(&bw, 3, 2) < 0) return 0;
    }
    if (zr_cstate_flush(&sml, &ct->ml, &bw) < 0) return 0;
    if (zr_cstate_flush(&sof, &ct->of, &bw) < 0) return 0;
    if (zr_cstate_flush(&sll, &ct->ll, &bw) < 0) return 0;
    const int bs = zr_bw_close(&bw);
    if (bs < 0) return 0;
    return zr_canon_close(dst, op + bs);
}

static inline int zr_canon_const4(const uint8_t *src, int n,
                                  uint8_t *dst, int dcap, const zr_ctables *ct)
{
    if (n < 3 || (int64_t)n * 4 >= (127 << 10)) return 0;
    const uint32_t content = (uint32_t)n * 4;
    int op = zr_canon_hdr(dst, dcap, content, 4);
    if (op < 0 || op + 40 > dcap) return 0;
    memcpy(dst + op, src, 4);
    op += 4;
    dst[op++] = 1;
    dst[op++] = 0x00;
    const uint32_t ml = (uint32_t)(n - 1) * 4;
    const uint8_t llc = 4, mlc = zr_ml_code(ml), ofc = 2;
    zr_bw bw;
    zr_bw_init(&bw, dst + op, dcap - op);
    zr_cstate sml, sof, sll;
    zr_cstate_init(&sml, &ct->ml, mlc);
    zr_cstate_init(&sof, &ct->of, ofc);
    zr_cstate_init(&sll, &ct->ll, llc);
    if (zr_bw_add(&bw, ml - ZR_ML_BASE[mlc], ZR_ML_BITS[mlc]) < 0) return 0;
    if (zr_bw_add(&bw, 3, 2) < 0) return 0;
    if (zr_cstate_flush(&sml, &ct->ml, &bw) < 0) return 0;
    if (zr_cstate_flush(&sof, &ct->of, &bw) < 0) return 0;
    if (zr_cstate_flush(&sll, &ct->ll, &bw) < 0) return 0;
    const int bs = zr_bw_close(&bw);
    if (bs < 0) return 0;
    return zr_canon_close(dst, op + bs);
}

#endif /* CSTRIPE_ZSTD_R_H */
