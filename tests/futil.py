"""Raw footer introspection for tests: parses the flat footer directory
(format.h layout) so tests can assert on-disk facts the public API hides
(segment count, stream-shape mode bytes, compressed sizes)."""
import struct

FOOTER_HEAD = struct.Struct("<IIIIQQbbHI")      # csf_footer_head (packed, 40 B)
COLDEF = struct.Struct("<32sBB6x")              # csf_coldef (40 B)
STRIPE_META = struct.Struct("<QQQQII")          # csf_stripe_meta (40 B)
SKIPNODE = struct.Struct("<qqQQQQQQIBBbBHH")    # csf_skipnode (76 B)
SEG = struct.Struct("<IIII")                    # csf_seg (16 B)

SEGMODE_GENERIC = 0x00
SEGMODE_P_BASE = 0x10
SEGMODE_CONST = 0x20
SEGMODE_LIT = 0x30
SEGMODE_ZR = 0x50
SEGMODE_ZRP_BASE = 0x60          # 0x60|L: canonical restricted-zstd P(L)
SEGMODE_ZR_CONST = 0x70
SEGMODE_ZR4B_BASE = 0x80         # 0x80|k: width-4 single-varying-byte slots
SEGMODE_ZR4_CONST = 0x84


def is_zr_mode(m):
    """any restricted-zstd stream shape (generic frames or canonical)"""
    return (m == SEGMODE_ZR or m == SEGMODE_ZR_CONST or 0x61 <= m <= 0x64 or
            0x80 <= m <= 0x84)


def read_footer(path):
    with open(path, "rb") as f:
        data = f.read()
    assert data[:8] == b"CSTRIPE1"
    assert data[-8:] == b"CSTRFOOT"
    (foff,) = struct.unpack("<Q", data[-16:-8])
    p = foff
    (version, n_cols, n_stripes, chunk_row_limit, stripe_row_limit,
     total_rows, compression, level, seg_kb, _res) = FOOTER_HEAD.unpack_from(data, p)
    p += FOOTER_HEAD.size
    cols = []
    for _ in range(n_cols):
        name, typ, scale = COLDEF.unpack_from(data, p)
        p += COLDEF.size
        cols.append({"name": name.split(b"\0")[0].decode(), "type": typ,
                     "scale": scale})
    stripes = []
    for _ in range(n_stripes):
        (file_offset, data_size, first_row, row_count,
         chunk_count, _r) = STRIPE_META.unpack_from(data, p)
        p += STRIPE_META.size
        group_rows = list(struct.unpack_from(f"<{chunk_count}I", data, p))
        p += 4 * chunk_count
        nodes = []
        for _c in range(n_cols):
            cn = []
            for _k in range(chunk_count):
                (min_i, max_i, row_cnt, value_off, value_len, exists_off,
                 exists_len, decomp_size, n_present, has_min_max, comp_type,
                 comp_level, _r1, n_segs, _r2) = SKIPNODE.unpack_from(data, p)
                p += SKIPNODE.size
                segs = []
                for _s in range(n_segs):
                    comp_off, comp_len, decomp_off, decomp_len = SEG.unpack_from(data, p)
                    p += SEG.size
                    segs.append({"comp_off": comp_off, "comp_len": comp_len,
                                 "decomp_off": decomp_off,
                                 "decomp_len": decomp_len & 0xFFFFFF,
                                 "mode": decomp_len >> 24})
                cn.append({"min_i": min_i, "max_i": max_i, "row_count": row_cnt,
                           "value_off": value_off, "value_len": value_len,
                           "decompressed_size": decomp_size,
                           "n_present": n_present, "has_min_max": has_min_max,
                           "comp_type": comp_type, "n_segs": n_segs,
                           "segs": segs,
                           "file_offset": file_offset})
            nodes.append(cn)
        stripes.append({"meta": {"file_offset": file_offset,
                                 "row_count": row_count,
                                 "chunk_count": chunk_count},
                        "group_rows": group_rows, "nodes": nodes})
    return {"version": version, "n_cols": n_cols, "cols": cols,
            "chunk_row_limit": chunk_row_limit, "total_rows": total_rows,
            "stripes": stripes}


def chunk_stream(path, node):
    """raw compressed bytes of one column-chunk's value stream"""
    with open(path, "rb") as f:
        f.seek(node["file_offset"] + node["value_off"])
        return f.read(node["value_len"])
