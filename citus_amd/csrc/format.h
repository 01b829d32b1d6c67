/*
 * format.h — on-disk stripe format, shared by writer and reader host code.
 *
 * Data layout follows the reference byte semantics:
 *  - per stripe, per column: all exists streams (chunk 0..n-1), then all
 *    value streams (chunk 0..n-1); offsets relative to the stripe's data
 *    start (FlushStripe, columnar_writer.c:391-516, esp. :425-458).
 *  - exists stream: bit-packed, bit i = row i present, LSB-first within a
 *    byte (SerializeBoolArray, columnar_writer.c:523-545); never compressed
 *    (SerializeChunkData :606-614 compresses only value buffers).
 *  - value stream: attlen-sized, att_align_nominal-aligned packed values of
 *    PRESENT rows only (SerializeSingleDatum, columnar_writer.c:555-585;
 *    DeserializeDatumArray, columnar_reader.c:1542-1572); for the fixed-width
 *    types supported here, a dense array of the non-null values.
 *  - value stream compression: whole-chunk LZ4/ZSTD (columnar_compression.c:
 *    62-158), except that a chunk MAY be stored as N independently decodable
 *    segments (each its own LZ4/ZSTD block over a slice of the decompressed
 *    stream, concatenated) with a segment table in the footer. N=1 is byte-
 *    compatible with the reference's whole-chunk block. Compressed-byte
 *    parity is unpinned by the reference's own tests;
 *    decompressed-content parity is what is pinned and checked.
 *
 * The footer replaces the columnar.stripe / chunk_group / chunk catalogs
 * (columnar_metadata.c:604-832; schema sql/columnar--9.5-1--10.0-1.sql:18-63)
 * with a flat directory, per SURVEY.md §2.
 *
 * All integers little-endian. This is a private format of this framework;
 * the drop-in boundary is include/cstripe.h, not these bytes.
 */
#ifndef CSTRIPE_FORMAT_H
#define CSTRIPE_FORMAT_H

#include <stdint.h>

#define CSF_MAGIC      "CSTRIPE1"      /* 8 bytes, file offset 0 */
#define CSF_FOOT_MAGIC "CSTRFOOT"      /* 8 bytes, end of file   */
#define CSF_HEADER_SIZE 16             /* magic + u32 version + u32 pad */
#define CSF_VERSION     1

/* footer tail: [u64 footer_offset][CSF_FOOT_MAGIC] — last 16 bytes */

/* ---- fixed-size footer records (packed, written/read as structs) ---- */

#pragma pack(push, 1)

typedef struct csf_footer_head {
    uint32_t version;
    uint32_t column_count;
    uint32_t stripe_count;
    uint32_t chunk_row_limit;      /* columnar.chunk_group_row_limit default 10000 */
    uint64_t stripe_row_limit;     /* columnar.stripe_row_limit default 150000 */
    uint64_t total_rows;
    uint8_t  compression;          /* requested codec (per-chunk actual in skip node) */
    int8_t   compression_level;
    uint16_t lz4_seg_target_kb;
    uint32_t reserved;
} csf_footer_head;

typedef struct csf_coldef {
    char     name[32];
    uint8_t  type;                 /* cstripe_type */
    uint8_t  scale;
    uint8_t  pad[6];
} csf_coldef;

typedef struct csf_stripe_meta {
    uint64_t file_offset;          /* absolute offset of stripe data start */
    uint64_t data_size;
    uint64_t first_row_number;
    uint64_t row_count;
    uint32_t chunk_count;
    uint32_t reserved;
} csf_stripe_meta;

/* mirror of ColumnChunkSkipNode (columnar.h:85-111) + our extensions
 * (n_present; segment table follows this record when n_segs > 1) */
typedef struct csf_skipnode {
    int64_t  min_i;                /* min/max in physical encoding; f64 via bit */
    int64_t  max_i;                /*   pattern (compare as double for F32/F64) */
    uint64_t row_count;            /* rows in chunk incl. nulls */
    uint64_t value_off;            /* relative to stripe data start */
    uint64_t value_len;            /* compressed length (sum of segments) */
    uint64_t exists_off;
    uint64_t exists_len;
    uint64_t decompressed_size;    /* raw value stream bytes */
    uint32_t n_present;            /* non-null rows (popcount of exists) */
    uint8_t  has_min_max;
    uint8_t  comp_type;            /* cstripe_compression actually applied */
    int8_t   comp_level;
    uint8_t  reserved;
    uint16_t n_segs;               /* >=1; seg table follows iff n_segs >= 1 */
    uint16_t reserved2;
} csf_skipnode;

/* one independently decodable compressed segment of a chunk's value stream.
 *
 * decomp_len bits 24-31 carry the segment's STREAM-SHAPE MODE, a redundant
 * hint the writer records about the standard-LZ4 stream it emitted (the
 * stream itself is always decodable by LZ4_decompress_safe, mode or not;
 * round-1 v1 files have 0 there since segments were < 16 MiB):
 *   0     generic         — arbitrary LZ4 block (greedy parse); sequence
 *                           parse required
 *   0x10|L canonical P(L) — width-8 values whose high (8-L) bytes are all
 *                           identical across the segment, emitted as the
 *                           periodic parse [lit v0+v1.low | off-8 match] then
 *                           [lit v(i).low | off-8 match]* then [lit v(n-1)];
 *                           every value's bytes sit at a CLOSED-FORM stream
 *                           position, so a GPU lane reads values straight out
 *                           of the compressed stream with no parsing
 *   0x20  constant        — all values equal v0: [lit v0 | off-8 match 8(n-2)]
 *                           [lit v(n-1)]; decode = broadcast v0
 *   0x30  literal         — one literal run; value i at hdr + i*width
 * Modes P/constant require 1 <= L <= 4 (match >= 4 bytes) and n >= 3 values. */
typedef struct csf_seg {
    uint32_t comp_off;             /* relative to chunk's value_off */
    uint32_t comp_len;
    uint32_t decomp_off;           /* relative to chunk's decompressed stream */
    uint32_t decomp_len;           /* bits 0-23 length; bits 24-31 mode */
} csf_seg;

#pragma pack(pop)

#define CSF_SEGMODE_GENERIC 0x00u
#define CSF_SEGMODE_P_BASE  0x10u  /* 0x10|L, L in 1..4 */
#define CSF_SEGMODE_CONST   0x20u
#define CSF_SEGMODE_LIT     0x30u
#define CSF_SEGMODE_ZR      0x50u  /* zstd: RESTRICTED frame this writer
                                    * emitted (raw literals + predefined-FSE
                                    * sequences, zstd_r.h) — device-decodable;
                                    * untagged zstd frames host-decode */
#define CSF_SEGMODE_ZRP_BASE 0x60u /* 0x60|L: CANONICAL restricted-zstd P(L)
                                    * frame (zr_canon_p): value 0's bytes at
                                    * stream offset 15, value j>=1's low-L
                                    * bytes at 23+(j-1)*L — closed-form GPU
                                    * access, still ZSTD_decompress-decodable */
#define CSF_SEGMODE_ZR_CONST 0x70u /* canonical restricted-zstd constant
                                    * (zr_canon_const): v0 at offset 15 */
#define CSF_SEGMODE_ZR4B_BASE 0x80u /* 0x80|k: width-4 slots differing only
                                     * in byte k (char(1) flag columns):
                                     * varying byte of row j at stream offset
                                     * 15 + (j==0 ? k : j+3+k); slot =
                                     * (slot0 & ~(0xFF<<8k)) | (b<<8k)
                                     * (zr_canon_b4) */
#define CSF_SEGMODE_ZR4_CONST 0x84u /* width-4 all-equal (zr_canon_const4):
                                     * slot0 at offset 15 */
#define CSF_SEG_DLEN_MASK   0x00FFFFFFu

static inline uint32_t csf_seg_dlen(const csf_seg *s) { return s->decomp_len & CSF_SEG_DLEN_MASK; }
static inline uint8_t  csf_seg_mode(const csf_seg *s) { return (uint8_t)(s->decomp_len >> 24); }

/* stream byte offset of value j's low-L bytes inside a canonical P(L)
 * segment (see layout above): j=0 -> 1, j=1 -> 9, else j*(L+3) + (6-L) */
static inline uint32_t csf_canon_p_pos(uint32_t j, uint32_t L)
{
    if (j == 0) return 1;
    if (j == 1) return 9;
    return j * (L + 3) + (6 - L);
}

/* stream byte offset of value j's low-L bytes inside a canonical
 * restricted-zstd P(L) frame (zr_canon_p layout): j=0 -> 15 (v0 stored
 * whole; its low-L bytes start there), else 23 + (j-1)*L */
static inline uint32_t csf_canon_zrp_pos(uint32_t j, uint32_t L)
{
    return j == 0 ? 15u : 23u + (j - 1u) * L;
}

/* stream byte offset of row j's varying byte inside a canonical width-4
 * single-varying-byte frame (zr_canon_b4 layout) */
static inline uint32_t csf_canon_zr4b_pos(uint32_t j, uint32_t k)
{
    return 15u + (j == 0 ? k : j + 3u + k);
}

/* C-collation (memcmp) order key of a short-varlena TEXT slot: the payload
 * chars big-endian, so integer order == byte-lexicographic order of the
 * text (zero padding ranks shorter strings first; PostgreSQL text cannot
 * contain NUL, so the key is injective). TEXT skip-node min/max store THIS
 * key, and refutation uses it only for the order-independent EQ/NE ops —
 * matching the reference's predicate_refuted_by pruning under "C"
 * collation for those operators (columnar_reader.c:1132-1187). */
static inline int64_t csf_text_lex_key(uint32_t slot)
{
    return (int64_t)((((slot >> 8) & 0xFFu) << 16) |
                     (((slot >> 16) & 0xFFu) << 8) |
                     ((slot >> 24) & 0xFFu));
}

/* literal-run header size for a mode-LIT segment of len decompressed bytes */
static inline uint32_t csf_canon_lit_hdr(uint32_t len)
{
    return len < 15 ? 1 : 2 + (len - 15) / 255;
}

/* footer layout, starting at footer_offset:
 *   csf_footer_head
 *   csf_coldef[column_count]
 *   for each stripe:
 *     csf_stripe_meta
 *     uint32 chunk_group_row_counts[chunk_count]
 *     for each column (0..column_count-1):
 *       for each chunk (0..chunk_count-1):
 *         csf_skipnode
 *         csf_seg[n_segs]          (always present, n_segs entries)
 */

static inline uint32_t csf_type_width(uint8_t t)
{
    switch (t) {
        case 1: return 1;          /* I8  */
        case 2: return 2;          /* I16 */
        case 3: return 4;          /* I32 */
        case 4: return 8;          /* I64 */
        case 5: return 4;          /* F32 */
        case 6: return 8;          /* F64 */
        case 7: return 4;          /* TEXT: short-varlena slot (hdr+<=3B) */
        default: return 0;
    }
}

#endif /* CSTRIPE_FORMAT_H */
