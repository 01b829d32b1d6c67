/*
 * pglz.h — PG-LZ codec restatement for COMPRESSION_PG_LZ chunks
 * (columnar_compression.c:122-151 compress, :230-262 decompress). Written
 * from the published pg_lzcompress format description (control byte of 8
 * LSB-first flags; flag=1 -> 2-3 byte tag: offset 12 bits, length 3..18
 * with a third byte extending 18+b; flag=0 -> one literal byte), not from
 * PostgreSQL sources.
 *
 * The stored chunk buffer carries the reference's ColumnarCompressHeader:
 * 4-byte varlena length word (total size << 2, "compressed" bit 0x02 on
 * little-endian) + int32 rawsize, then the PG-LZ byte stream
 * (columnar_compression.c:38-53).
 *
 * Shared by the host reader (staging/next_batch decode — pglz is a
 * host-side codec here: reference-migrated tables read correctly; the GPU
 * codecs are LZ4 (device) and ZSTD) and by the oracle.
 */
#ifndef CSTRIPE_PGLZ_H
#define CSTRIPE_PGLZ_H

#include <stdint.h>
#include <string.h>

#define CSPGLZ_HDRSZ 8

static inline uint32_t cspglz_varsize(const uint8_t *buf)
{
    uint32_t w;
    memcpy(&w, buf, 4);
    return w >> 2;                     /* 4-byte little-endian varlena */
}

static inline int32_t cspglz_rawsize(const uint8_t *buf)
{
    int32_t r;
    memcpy(&r, buf + 4, 4);
    return r;
}

static inline void cspglz_set_header(uint8_t *buf, uint32_t total, int32_t rawsize)
{
    uint32_t w = (total << 2) | 0x02;  /* SET_VARSIZE_COMPRESSED */
    memcpy(buf, &w, 4);
    memcpy(buf + 4, &rawsize, 4);
}

/* decompress the PG-LZ byte stream (no header) into dst; returns bytes
 * produced or -1 on corruption. Mirrors pglz_decompress(.., check=true)
 * semantics: exact dlen required. */
static inline int32_t cspglz_decompress(const uint8_t *src, int32_t slen,
                                        uint8_t *dst, int32_t dlen)
{
    int32_t ip = 0, op = 0;
    while (ip < slen) {
        uint8_t ctrl = src[ip++];
        for (int bit = 0; bit < 8 && ip < slen; bit++, ctrl >>= 1) {
            if (ctrl & 1) {
                if (ip + 2 > slen) return -1;
                int32_t len = (src[ip] & 0x0F) + 3;
                int32_t off = ((src[ip] & 0xF0) << 4) | src[ip + 1];
                ip += 2;
                if (len == 18) {
                    if (ip >= slen) return -1;
                    len += src[ip++];
                }
                if (off == 0 || off > op || op + len > dlen) return -1;
                /* overlapping copy: byte-by-byte forward (RLE semantics) */
                for (int32_t j = 0; j < len; j++, op++)
                    dst[op] = dst[op - off];
            } else {
                if (op >= dlen) return -1;
                dst[op++] = src[ip++];
            }
        }
    }
    return op == dlen ? op : -1;
}

/* minimal greedy PG-LZ compressor (hash of 3-byte prefixes, 4 KB history
 * like the format's 12-bit offsets allow). Used by the writer for
 * COMPRESSION_PG_LZ chunks and by the decoder's differential tests.
 * Returns compressed size (stream only, no header) or 0 if dst too small /
 * stream would not shrink. */
#define CSPGLZ_HASH_SIZE 4096

static inline uint32_t cspglz_hash(const uint8_t *p)
{
    return ((uint32_t)p[0] << 6 ^ (uint32_t)p[1] << 3 ^ (uint32_t)p[2]) &
           (CSPGLZ_HASH_SIZE - 1);
}

static inline int32_t cspglz_compress(const uint8_t *src, int32_t slen,
                                      uint8_t *dst, int32_t dcap)
{
    int32_t table[CSPGLZ_HASH_SIZE];
    for (int i = 0; i < CSPGLZ_HASH_SIZE; i++) table[i] = -1;
    int32_t ip = 0, op = 0;
    while (ip < slen) {
        if (op + 1 + 8 * 3 > dcap) return 0;       /* worst-case group */
        int32_t ctrl_pos = op++;
        uint8_t ctrl = 0;
        for (int bit = 0; bit < 8 && ip < slen; bit++) {
            int32_t mpos = -1, mlen = 0;
            if (ip + 3 <= slen) {
                uint32_t h = cspglz_hash(src + ip);
                int32_t cand = table[h];
                table[h] = ip;
                if (cand >= 0 && ip - cand <= 0x0FFF &&
                    memcmp(src + cand, src + ip, 3) == 0) {
                    int32_t maxlen = slen - ip;
                    if (maxlen > 273) maxlen = 273;   /* 18 + 255 */
                    int32_t l = 3;
                    while (l < maxlen && src[cand + l] == src[ip + l]) l++;
                    mpos = cand;
                    mlen = l;
                }
            }
            if (mlen >= 3) {
                int32_t off = ip - mpos;
                int32_t lenfld = mlen < 18 ? mlen : 18;
                dst[op++] = (uint8_t)(((lenfld - 3) & 0x0F) | ((off >> 4) & 0xF0));
                dst[op++] = (uint8_t)(off & 0xFF);
                if (lenfld == 18)
                    dst[op++] = (uint8_t)(mlen - 18);
                ctrl |= (uint8_t)(1 << bit);
                ip += mlen;
            } else {
                dst[op++] = src[ip++];
            }
        }
        dst[ctrl_pos] = ctrl;
    }
    return op < slen ? op : 0;
}

#endif
