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
 * ratio cost (the output is standard LZ4 either way). */
static inline int lz4e_compress_mm(lz4e_state *st, const uint8_t *src, int slen,
                                   uint8_t *dst, int dcap, int min_match)
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
            uint32_t cur32 = lz4e_read32(src + pos);
            uint32_t h = lz4e_hash(cur32);
            uint32_t cand = st->table[h];
            st->table[h] = base + (uint32_t)pos;
            int mpos = (int)(cand - base);       /* may be negative/garbage */
            if (cand >= base && mpos < pos && pos - mpos <= 65535 &&
                lz4e_read32(src + mpos) == cur32) {
                /* extend match */
                int mlen = 4;
                while (pos + mlen < matchlimit && src[mpos + mlen] == src[pos + mlen])
                    mlen++;
                if (mlen < min_match) { pos++; continue; }
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

static inline int lz4e_compress(lz4e_state *st, const uint8_t *src, int slen,
                                uint8_t *dst, int dcap)
{
    return lz4e_compress_mm(st, src, slen, dst, dcap, 4);
}

#endif
