/*
 * lz4_enc.h — minimal LZ4 block-format encoder for micro-segments.
 *
 * Why not liblz4: LZ4_compress_default resets a 16 KiB hash state per call,
 * which is ~64x the payload for this writer's 256 B parallel-decode
 * segments. This encoder keeps a persistent 4096-entry u32 table with a
 * monotonically increasing base, so repeated calls need no reset.
 *
 * Output is standard LZ4 block format (token / litlen ext / literals /
 * little-endian 16-bit offset / matchlen ext; >=4-byte matches; last 5
 * bytes literal; no match starting within the last 12) — decodable by
 * LZ4_decompress_safe, which the oracle and the format tests use to pin
 * compatibility (the reference consumes the same block format,
 * columnar_compression.c:183). Greedy parse; written from the public block
 * format specification, not from lz4 sources.
 */
#ifndef CSTRIPE_LZ4_ENC_H
#define CSTRIPE_LZ4_ENC_H

#include <stdint.h>
#include <string.h>

#define LZ4E_TABLE_SIZE 4096

typedef struct lz4e_state {
    uint32_t table[LZ4E_TABLE_SIZE];   /* absolute position + 1 (0 = empty) */
    uint32_t base;                     /* absolute position of current src[0] */
} lz4e_state;

static inline void lz4e_init(lz4e_state *st)
{
    memset(st->table, 0, sizeof(st->table));
    st->base = 1;                      /* keep entry 0 meaning "empty" */
}

static inline uint32_t lz4e_hash(uint32_t v)
{
    return (v * 2654435761u) >> (32 - 12);
}

static inline uint32_t lz4e_read32(const uint8_t *p)
{
    uint32_t v;
    memcpy(&v, p, 4);
    return v;
}

/* returns compressed size, or 0 if dst capacity insufficient.
 * min_match (>= 4): matches shorter than this are skipped — fewer, longer
 * sequences decode much faster on the lane-parallel GPU decoder at a small
 * ratio cost (the output is standard LZ4 either way).
 * align (1, 2 or 4): restrict match positions/offsets/lengths to multiples
 * of the column's value width, so every copy in the stream is word-granular
 * — the GPU lane decoder then runs its aligned-word path throughout (the
 * dominant win for 4-byte varlena slot streams). Standard LZ4 either way. */
static inline int lz4e_compress_mm_a(lz4e_state *st, const uint8_t *src, int slen,
                                     uint8_t *dst, int dcap, int min_match,
                                     int align)
{
    const uint32_t base = st->base;
    st->base += (uint32_t)slen + 1;
    if (st->base < base) lz4e_init(st);          /* absolute counter wrap */

    uint8_t *op = dst;
    uint8_t *const oend = dst + dcap;
    int anchor = 0;

    if (slen >= 13) {
        const int mflimit = slen - 12;           /* no match starts after this */
        const int matchlimit = slen - 5;         /* matches may extend to here */
        int pos = 0;
        while (pos <= mflimit) {
            if (align > 1 && (pos % align) != 0) { pos++; continue; }
            uint32_t cur32 = lz4e_read32(src + pos);
            uint32_t h = lz4e_hash(cur32);
            uint32_t cand = st->table[h];
            st->table[h] = base + (uint32_t)pos;
            int mpos = (int)(cand - base);       /* may be negative/garbage */
            if (cand >= base && mpos < pos && pos - mpos <= 65535 &&
                (align <= 1 || ((pos - mpos) % align) == 0) &&
                lz4e_read32(src + mpos) == cur32) {
                /* extend match */
                int mlen = 4;
                while (pos + mlen < matchlimit && src[mpos + mlen] == src[pos + mlen])
                    mlen++;
                if (align > 1) mlen -= mlen % align;
                if (mlen < min_match || mlen < 4) { pos++; continue; }
                int litlen = pos - anchor;
                /* emit: token + ext + literals + offset + ext */
                uint8_t *tok = op++;
                if (op >= oend) return 0;
                if (litlen >= 15) {
                    int l = litlen - 15;
                    while (l >= 255) { if (op >= oend) return 0; *op++ = 255; l -= 255; }
                    if (op >= oend) return 0;
                    *op++ = (uint8_t)l;
                }
                if (op + litlen + 2 > oend) return 0;
                memcpy(op, src + anchor, (size_t)litlen);
                op += litlen;
                uint16_t off = (uint16_t)(pos - mpos);
                *op++ = (uint8_t)off;
                *op++ = (uint8_t)(off >> 8);
                int mex = mlen - 4;
                if (mex >= 15) {
                    *tok = (uint8_t)((litlen >= 15 ? 15 : litlen) << 4 | 15);
                    mex -= 15;
                    while (mex >= 255) { if (op >= oend) return 0; *op++ = 255; mex -= 255; }
                    if (op >= oend) return 0;
                    *op++ = (uint8_t)mex;
                } else {
                    *tok = (uint8_t)((litlen >= 15 ? 15 : litlen) << 4 | mex);
                }
                pos += mlen;
                anchor = pos;
            } else {
                pos++;
            }
        }
    }

    /* final literals */
    {
        int litlen = slen - anchor;
        uint8_t *tok = op++;
        if (op > oend) return 0;
        if (litlen >= 15) {
            int l = litlen - 15;
            *tok = (uint8_t)(15 << 4);
            while (l >= 255) { if (op >= oend) return 0; *op++ = 255; l -= 255; }
            if (op >= oend) return 0;
            *op++ = (uint8_t)l;
        } else {
            *tok = (uint8_t)(litlen << 4);
        }
        if (op + litlen > oend) return 0;
        memcpy(op, src + anchor, (size_t)litlen);
        op += litlen;
    }
    return (int)(op - dst);
}

static inline int lz4e_compress_mm(lz4e_state *st, const uint8_t *src, int slen,
                                   uint8_t *dst, int dcap, int min_match)
{
    return lz4e_compress_mm_a(st, src, slen, dst, dcap, min_match, 1);
}

static inline int lz4e_compress(lz4e_state *st, const uint8_t *src, int slen,
                                uint8_t *dst, int dcap)
{
    return lz4e_compress_mm(st, src, slen, dst, dcap, 4);
}

/* =====================================================================
 * Canonical emitters — alternative PARSES of the same input that are still
 * standard LZ4 blocks (decodable by LZ4_decompress_safe, pinned in
 * tests/test_format.py) but whose byte layout is closed-form, so a GPU
 * lane can read any value straight out of the compressed stream without
 * walking the sequence chain. See format.h CSF_SEGMODE_* for the layout.
 * ===================================================================== */

/* canonical P(L): n width-8 values, all sharing their high (8-L) bytes
 * (1 <= L <= 4, n >= 3). Emits:
 *   S0:       [token lit=8+L match=8-L] v0(8B) v1.low(L) [off=8]
 *   S1..Sn-3: [token lit=L   match=8-L] v(i+1).low(L)    [off=8]
 *   final:    [token lit=8] v(n-1)(8B)
 * (last match starts 16-L >= 12 bytes before end; trailing 8 literals
 * satisfy the 5-literal end rule — LZ4 block format constraints) */
static inline int lz4e_canon_p(const uint8_t *src, int n, int L,
                               uint8_t *dst, int dcap)
{
    if (n < 3 || L < 1 || L > 4) return 0;
    const int need = (11 + L) + (n - 3) * (L + 3) + 9;
    if (need > dcap) return 0;
    uint8_t *op = dst;
    const uint8_t mtok = (uint8_t)(8 - L - 4);
    *op++ = (uint8_t)(((8 + L) << 4) | mtok);
    memcpy(op, src, 8 + (size_t)L);          /* v0 full + v1 low */
    op += 8 + L;
    *op++ = 8; *op++ = 0;                    /* offset 8 */
    for (int i = 1; i <= n - 3; i++) {
        *op++ = (uint8_t)((L << 4) | mtok);
        memcpy(op, src + (size_t)(i + 1) * 8, (size_t)L);
        op += L;
        *op++ = 8; *op++ = 0;
    }
    *op++ = (uint8_t)(8 << 4);               /* final: 8 literals */
    memcpy(op, src + (size_t)(n - 1) * 8, 8);
    op += 8;
    return (int)(op - dst);
}

/* canonical constant: n equal width-8 values (n >= 3):
 *   S0: [token lit=8 match=ext] v0(8B) [off=8] [ext...]   (match 8(n-2) B)
 *   final: [token lit=8] v0(8B) */
static inline int lz4e_canon_const(const uint8_t *src, int n,
                                   uint8_t *dst, int dcap)
{
    if (n < 3) return 0;
    int mlen = 8 * (n - 2);                  /* bytes the match reproduces */
    int mex = mlen - 4;
    uint8_t *op = dst;
    uint8_t *oend = dst + dcap;
    if (op + 11 > oend) return 0;
    *op++ = (uint8_t)((8 << 4) | (mex >= 15 ? 15 : mex));
    memcpy(op, src, 8);
    op += 8;
    *op++ = 8; *op++ = 0;
    if (mex >= 15) {
        mex -= 15;
        while (mex >= 255) { if (op >= oend) return 0; *op++ = 255; mex -= 255; }
        if (op >= oend) return 0;
        *op++ = (uint8_t)mex;
    }
    if (op + 9 > oend) return 0;
    *op++ = (uint8_t)(8 << 4);
    memcpy(op, src, 8);
    op += 8;
    return (int)(op - dst);
}

/* canonical literal run: the whole stream as one literal sequence
 * (value i at csf_canon_lit_hdr(slen) + i*width) */
static inline int lz4e_canon_lit(const uint8_t *src, int slen,
                                 uint8_t *dst, int dcap)
{
    uint8_t *op = dst;
    int hdr = slen < 15 ? 1 : 2 + (slen - 15) / 255;
    if (hdr + slen > dcap) return 0;
    if (slen < 15) {
        *op++ = (uint8_t)(slen << 4);
    } else {
        *op++ = (uint8_t)(15 << 4);
        int l = slen - 15;
        while (l >= 255) { *op++ = 255; l -= 255; }
        *op++ = (uint8_t)l;
    }
    memcpy(op, src, (size_t)slen);
    return hdr + slen;
}

#endif
