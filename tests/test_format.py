"""
Format/writer tests: round trips through the product writer and the oracle
reader; reference-compatibility of the LZ4 chunk blocks (a 1-segment chunk
is exactly the whole-chunk block DecompressBuffer consumes — decoded here
with the SAME system liblz4 call the reference uses).
"""
import ctypes as C
import os
import struct

import numpy as np
import pytest

import citus_amd as ca
import oracle

RNG = np.random.default_rng(42)


def roundtrip(tmp_path, defs, cols, nulls=None, **kw):
    path = str(tmp_path / "t.cs")
    ca.write_table(path, defs, [np.ascontiguousarray(c) for c in cols],
                   nulls=nulls, **kw)
    t = oracle.OracleTable(path)
    return path, t


def read_all(t, col, n_rows, dtype, chunk_rows, stripe_rows=10000):
    vals = np.zeros(chunk_rows, dtype=dtype)
    ex = np.zeros(chunk_rows, dtype=np.uint8)
    out_v, out_e = [], []
    got = 0
    stripe = 0
    while got < n_rows:
        in_stripe = min(stripe_rows, n_rows - got)
        chunk = 0
        sdone = 0
        while sdone < in_stripe:
            take = min(chunk_rows, in_stripe - sdone)
            t.read_chunk(stripe, chunk, col, vals, ex)
            out_v.append(vals[:take].copy())
            out_e.append(ex[:take].copy())
            sdone += take
            chunk += 1
        got += in_stripe
        stripe += 1
    return np.concatenate(out_v), np.concatenate(out_e)


@pytest.mark.parametrize("comp", [ca.COMP_NONE, ca.COMP_LZ4, ca.COMP_ZSTD])
def test_roundtrip_types(tmp_path, comp):
    n = 25000
    defs = [("a", ca.I64, 0), ("b", ca.I32, 0), ("c", ca.I16, 0),
            ("d", ca.I8, 0), ("e", ca.F64, 0), ("f", ca.F32, 0)]
    cols = [RNG.integers(-2**62, 2**62, n).astype(np.int64),
            RNG.integers(-2**31, 2**31 - 1, n).astype(np.int32),
            RNG.integers(-2**15, 2**15 - 1, n).astype(np.int16),
            RNG.integers(-128, 127, n).astype(np.int8),
            RNG.normal(size=n),
            RNG.normal(size=n).astype(np.float32)]
    path, t = roundtrip(tmp_path, defs, cols, compression=comp,
                        stripe_row_limit=10000, chunk_group_row_limit=3000)
    assert t.row_count == n
    dtypes = [np.int64, np.int32, np.int16, np.int8, np.float64, np.float32]
    for ci, (col, dt) in enumerate(zip(cols, dtypes)):
        v, e = read_all(t, ci, n, dt, 3000)
        assert e.all()
        np.testing.assert_array_equal(v, col)
    t.close()


def test_roundtrip_nulls(tmp_path):
    n = 7777
    a = RNG.integers(0, 1000, n).astype(np.int64)
    na = (RNG.random(n) < 0.3).astype(np.uint8)
    path, t = roundtrip(tmp_path, [("a", ca.I64, 0)], [a], nulls=[na],
                        compression=ca.COMP_LZ4, chunk_group_row_limit=1000)
    v, e = read_all(t, 0, n, np.int64, 1000)
    np.testing.assert_array_equal(e, 1 - na)
    np.testing.assert_array_equal(v[na == 0], a[na == 0])
    assert (v[na == 1] == 0).all()           # null slots zero-filled
    t.close()


def test_edge_sizes(tmp_path):
    for n in (1, 999, 1000, 1001, 10000, 10001):
        a = np.arange(n, dtype=np.int64)
        path = str(tmp_path / f"e{n}.cs")
        ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_LZ4,
                       stripe_row_limit=10000, chunk_group_row_limit=1000)
        with oracle.OracleTable(path) as t:
            assert t.row_count == n
            parts, _ = t.scan_agg([], [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1),
                                       (ca.AGG_MIN_I64, 0), (ca.AGG_MAX_I64, 0)])
            assert parts[0].i128 == n * (n - 1) // 2
            assert parts[1].count == n
            assert parts[2].i128 == 0
            assert parts[3].i128 == n - 1


def test_int64_extremes(tmp_path):
    a = np.array([np.iinfo(np.int64).min, -1, 0, 1, np.iinfo(np.int64).max],
                 dtype=np.int64)
    path = str(tmp_path / "x.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_LZ4)
    with oracle.OracleTable(path) as t:
        parts, _ = t.scan_agg([], [(ca.AGG_MIN_I64, 0), (ca.AGG_MAX_I64, 0),
                                   (ca.AGG_SUM_I64, 0)])
        assert parts[0].i128 == np.iinfo(np.int64).min
        assert parts[1].i128 == np.iinfo(np.int64).max
        assert parts[2].i128 == -1


def test_empty_table(tmp_path):
    path = str(tmp_path / "empty.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [np.array([], dtype=np.int64)],
                   compression=ca.COMP_LZ4)
    with oracle.OracleTable(path) as t:
        assert t.row_count == 0
        parts, _ = t.scan_agg([], [(ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_I64, 0)])
        assert parts[0].count == 0 and not parts[0].is_null   # COUNT coalesces to 0
        assert parts[1].is_null                               # strict SUM -> NULL


def test_single_segment_block_is_reference_decodable(tmp_path):
    """With a huge segment target a chunk is ONE LZ4 block — byte-decodable by
    the very LZ4_decompress_safe call DecompressBuffer makes
    (columnar_compression.c:183). Pins the on-disk compatibility claim."""
    n = 10000
    a = (np.arange(n, dtype=np.int64) % 97)
    path = str(tmp_path / "one.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_LZ4,
                   lz4_seg_target_kb=1024)  # 80 KB chunk << 1 MB -> 1 segment

    # parse the footer by hand to find the chunk's compressed bytes
    raw = open(path, "rb").read()
    assert raw[:8] == b"CSTRIPE1" and raw[-8:] == b"CSTRFOOT"
    (foff,) = struct.unpack("<Q", raw[-16:-8])
    HEAD, SMETA, NODE = "<IIIIQQbbHI", "<QQQQII", "<qqQQQQQQIBBbBHH"
    p = foff
    head = struct.unpack(HEAD, raw[p:p + struct.calcsize(HEAD)])
    p += struct.calcsize(HEAD) + head[1] * 40    # skip coldefs
    smeta = struct.unpack(SMETA, raw[p:p + struct.calcsize(SMETA)])
    p += struct.calcsize(SMETA) + smeta[4] * 4   # skip chunk_group_row_counts
    node = struct.unpack(NODE, raw[p:p + struct.calcsize(NODE)])
    (mn, mx, rowcnt, voff, vlen, eoff, elen, dsize, npres,
     hasmm, ctype, clevel, _r, nsegs, _r2) = node
    assert nsegs == 1 and ctype == ca.COMP_LZ4
    comp = raw[smeta[0] + voff: smeta[0] + voff + vlen]

    lz4 = C.CDLL("liblz4.so.1")
    out = C.create_string_buffer(int(dsize))
    r = lz4.LZ4_decompress_safe(comp, out, len(comp), int(dsize))
    assert r == dsize
    np.testing.assert_array_equal(np.frombuffer(out.raw, dtype=np.int64), a)


def test_exists_bitpacking_matches_reference_layout(tmp_path):
    """SerializeBoolArray: bit i -> byte i/8, bit position i%8 (LSB first)."""
    n = 20
    a = np.arange(n, dtype=np.int64)
    nl = np.zeros(n, dtype=np.uint8)
    nl[[1, 3, 8, 15]] = 1
    path = str(tmp_path / "bits.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [a], nulls=[nl],
                   compression=ca.COMP_NONE)
    raw = open(path, "rb").read()
    (foff,) = struct.unpack("<Q", raw[-16:-8])
    HEAD, SMETA, NODE = "<IIIIQQbbHI", "<QQQQII", "<qqQQQQQQIBBbBHH"
    p = foff + struct.calcsize(HEAD) + 40        # head + 1 coldef
    smeta = struct.unpack(SMETA, raw[p:p + struct.calcsize(SMETA)])
    p += struct.calcsize(SMETA) + smeta[4] * 4
    node = struct.unpack(NODE, raw[p:p + struct.calcsize(NODE)])
    eoff, elen = node[5], node[6]
    bits = raw[smeta[0] + eoff: smeta[0] + eoff + elen]
    expect = bytearray((n + 7) // 8)
    for i in range(n):
        if not nl[i]:
            expect[i // 8] |= 1 << (i % 8)
    assert bits == bytes(expect)


def test_corrupted_footer_fails_cleanly(tmp_path):
    """Truncated/corrupted files: clean error, no crash (the ABI's ereport
    analog)."""
    import citus_amd as ca2
    path = str(tmp_path / "c.cs")
    ca2.gen_lineitem(path, 10_000)
    raw = bytearray(open(path, "rb").read())
    # corrupt footer magic
    bad1 = str(tmp_path / "bad1.cs")
    b = bytearray(raw)
    b[-4:] = b"XXXX"
    open(bad1, "wb").write(bytes(b))
    with pytest.raises(ca2.CStripeError):
        ca2.Reader(bad1)
    # truncate mid-file
    bad2 = str(tmp_path / "bad2.cs")
    open(bad2, "wb").write(bytes(raw[:len(raw) // 3]))
    with pytest.raises(ca2.CStripeError):
        ca2.Reader(bad2)
    # absurd footer offset
    bad3 = str(tmp_path / "bad3.cs")
    b = bytearray(raw)
    b[-16:-8] = (2**60).to_bytes(8, "little")
    open(bad3, "wb").write(bytes(b))
    with pytest.raises(ca2.CStripeError):
        ca2.Reader(bad3)


def test_tail_segment_not_absorbed(tmp_path):
    """Round-1 advisor (high): a dense i64 chunk with rows % 32 == 1 used to
    absorb the 8 B tail into the final 256 B segment (257-271 B), which the
    fused kernels then read from the wrong lane region. The writer now emits
    the tiny tail as its own segment; the stream must still round-trip."""
    for n in (321, 10001, 9985):
        a = (np.arange(n, dtype=np.int64) * 7) % 5000
        path = str(tmp_path / f"tail{n}.cs")
        ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_LZ4)
        with oracle.OracleTable(path) as t:
            parts, _ = t.scan_agg([], [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1)])
            assert parts[0].i128 == int(a.sum())
            assert parts[1].count == n


def test_n_segs_over_cap_rejected(tmp_path):
    """Round-1 advisor (medium): a chunk needing > 4096 segments used to be
    silently truncated to uint16 / misparsed; the writer must now refuse."""
    n = 10000  # 80 KB i64 chunk at 16 B/segment = 5000 segments > 4096
    a = RNG.integers(0, 127, n).astype(np.int64)   # compressible (stays LZ4)
    path = str(tmp_path / "overcap.cs")
    with pytest.raises(ca.CStripeError, match="4096"):
        ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_LZ4,
                       lz4_seg_target_bytes=16, canonical=0)


def test_shard_chunk_row_limit_mismatch_rejected(tmp_path):
    """Round-1 advisor (medium): shard files with a larger chunk_group_row_limit
    than file 0's had rows silently dropped; the directory open must reject."""
    d = tmp_path / "shards"
    d.mkdir()
    a = np.arange(5000, dtype=np.int64)
    ca.write_table(str(d / "a.cs"), [("a", ca.I64, 0)], [a],
                   compression=ca.COMP_LZ4, chunk_group_row_limit=1000)
    ca.write_table(str(d / "b.cs"), [("a", ca.I64, 0)], [a],
                   compression=ca.COMP_LZ4, chunk_group_row_limit=2000)
    r = ca._open(str(d).encode())
    assert not r, "mismatched chunk_row_limit shard dir must fail to open"
    assert "chunk_group_row_limit" in ca.errmsg()


def test_nan_pruning_not_refuted(tmp_path):
    """Round-1 advisor (low): PG float ordering treats NaN as greater than
    every value, so a chunk whose only rows matching `col > c` are NaN must
    NOT be pruned, and `col > c` must match NaN rows (float8_gt semantics)."""
    n = 1000
    a = np.full(n, 5.0)
    a[::10] = np.nan          # chunk min/max: finite 5.0 .. NaN (PG order)
    path = str(tmp_path / "nan.cs")
    ca.write_table(path, [("a", ca.F64, 0)], [a], compression=ca.COMP_LZ4,
                   chunk_group_row_limit=500)
    with oracle.OracleTable(path) as t:
        # col > 7: only NaN rows qualify; chunk must survive and count 100
        parts, filtered = t.scan_agg([(0, ca.PRED_GT, 7.0)],
                                     [(ca.AGG_COUNT_STAR, -1)])
        assert filtered == 0
        assert parts[0].count == n // 10
        # col < 7: NaN rows excluded (NaN sorts high)
        parts, _ = t.scan_agg([(0, ca.PRED_LT, 7.0)], [(ca.AGG_COUNT_STAR, -1)])
        assert parts[0].count == n - n // 10
        # MAX over a NaN-containing column is NaN; MIN is the finite value
        parts, _ = t.scan_agg([], [(ca.AGG_MAX_F64, 0), (ca.AGG_MIN_F64, 0)])
        assert np.isnan(parts[0].f64)
        assert parts[1].f64 == 5.0


def test_all_nan_min_is_nan(tmp_path):
    a = np.full(64, np.nan)
    path = str(tmp_path / "allnan.cs")
    ca.write_table(path, [("a", ca.F64, 0)], [a], compression=ca.COMP_LZ4)
    with oracle.OracleTable(path) as t:
        parts, _ = t.scan_agg([], [(ca.AGG_MIN_F64, 0), (ca.AGG_MAX_F64, 0)])
        assert np.isnan(parts[0].f64) and np.isnan(parts[1].f64)
        assert parts[0].count == 64


# ---------- canonical stream-shape modes (round 2; format.h) ----------

import futil


def _liblz4():
    lz4 = C.CDLL("liblz4.so.1")
    lz4.LZ4_decompress_safe.restype = C.c_int
    lz4.LZ4_decompress_safe.argtypes = [C.c_char_p, C.c_char_p, C.c_int, C.c_int]
    return lz4


def _decode_sys(lz4, comp, dlen):
    out = C.create_string_buffer(dlen)
    r = lz4.LZ4_decompress_safe(bytes(comp), out, len(comp), dlen)
    assert r == dlen, f"system LZ4_decompress_safe returned {r}, want {dlen}"
    return out.raw


@pytest.mark.parametrize("case", [
    ("L1", lambda n: np.arange(n, dtype=np.int64) % 200),
    ("L2", lambda n: 1000 + (np.arange(n, dtype=np.int64) * 37) % 50000),
    ("L3", lambda n: (np.arange(n, dtype=np.int64) * 104729) % (1 << 23)),
    ("L4", lambda n: (np.arange(n, dtype=np.int64) * 2654435761) % (1 << 31)),
    ("neg", lambda n: -((np.arange(n, dtype=np.int64) * 13) % 100) - 1),
    ("const", lambda n: np.full(n, -123456789, dtype=np.int64)),
    # f64 whose low mantissa bytes vary under a fixed exponent: the bit
    # patterns share their high bytes, exactly the canonical-P shape
    ("f64", lambda n: (np.uint64(0x3FF0000000000000)
                       | (np.arange(n, dtype=np.uint64) % 251)).view(np.float64)),
])
@pytest.mark.parametrize("n", [3, 4, 37, 9999, 10000])
def test_canonical_streams_system_decodable(tmp_path, case, n):
    """The canonical parses (P(L)/const) are STANDARD LZ4 blocks: the system
    LZ4_decompress_safe — the exact call the reference's DecompressBuffer
    makes (columnar_compression.c:183) — must reproduce the raw value stream
    byte-for-byte, and the footer must carry the expected mode tag."""
    name, gen = case
    a = np.ascontiguousarray(gen(n))
    typ = ca.F64 if a.dtype == np.float64 else ca.I64
    path = str(tmp_path / f"c_{name}_{n}.cs")
    ca.write_table(path, [("a", typ, 0)], [a], compression=ca.COMP_LZ4)
    foot = futil.read_footer(path)
    node = foot["stripes"][0]["nodes"][0][0]
    if n <= 4:
        # tiny chunks may store raw (no shrink; reference CompressBuffer
        # rule) or pick the greedy parse; only content parity matters
        with oracle.OracleTable(path) as t:
            kind = ca.AGG_SUM_F64 if typ == ca.F64 else ca.AGG_SUM_I64
            parts, _ = t.scan_agg([], [(kind, 0)])
            if typ == ca.F64:
                assert abs(parts[0].f64 - float(a.sum())) < 1e-9
            else:
                assert parts[0].i128 == int(a.sum())
        return
    assert node["comp_type"] == ca.COMP_LZ4
    assert node["n_segs"] == 1
    seg = node["segs"][0]
    if name == "const":
        assert seg["mode"] == futil.SEGMODE_CONST
    else:
        assert futil.SEGMODE_P_BASE < seg["mode"] <= futil.SEGMODE_P_BASE + 4, \
            f"expected P mode, got {seg['mode']:#x}"
    assert seg["decomp_len"] == node["decompressed_size"]
    comp = futil.chunk_stream(path, node)
    raw = _decode_sys(_liblz4(), comp, seg["decomp_len"])
    assert raw == a.tobytes()
    # compression must actually compress for these shapes
    assert len(comp) < len(raw)


def test_canonical_mode_choice(tmp_path):
    """Writer mode decisions: high-entropy 8B values (L>4) fall back to the
    greedy parse (or raw when incompressible, reference CompressBuffer
    semantics); sub-8B widths stay greedy; canonical=0 disables tagging."""
    n = 10000
    rnd = RNG.integers(-2**62, 2**62, n).astype(np.int64)      # incompressible
    lowcard = RNG.integers(0, 10, n).astype(np.int64)          # canonical P(1)
    flags = RNG.integers(0, 3, n).astype(np.int8)              # i8 -> greedy
    path = str(tmp_path / "mix.cs")
    ca.write_table(path, [("r", ca.I64, 0), ("l", ca.I64, 0), ("f", ca.I8, 0)],
                   [rnd, lowcard, flags], compression=ca.COMP_LZ4)
    foot = futil.read_footer(path)
    nodes = foot["stripes"][0]["nodes"]
    assert nodes[0][0]["comp_type"] == ca.COMP_NONE             # incompressible
    assert nodes[1][0]["segs"][0]["mode"] == futil.SEGMODE_P_BASE + 1
    assert nodes[2][0]["comp_type"] == ca.COMP_LZ4
    assert all(s["mode"] == futil.SEGMODE_GENERIC for s in nodes[2][0]["segs"])

    path2 = str(tmp_path / "off.cs")
    ca.write_table(path2, [("l", ca.I64, 0)], [lowcard],
                   compression=ca.COMP_LZ4, canonical=0)
    foot2 = futil.read_footer(path2)
    assert all(s["mode"] == futil.SEGMODE_GENERIC
               for s in foot2["stripes"][0]["nodes"][0][0]["segs"])


def test_canonical_sparse_roundtrip(tmp_path):
    """Canonical addressing is by VALUE index, so NULL-bearing chunks work:
    present values round-trip, null slots read as null (oracle reader)."""
    n = 10000
    a = (np.arange(n, dtype=np.int64) * 7) % 3000
    nulls = (np.arange(n) % 5 == 0).astype(np.uint8)
    path = str(tmp_path / "sp.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [a], nulls=[nulls],
                   compression=ca.COMP_LZ4)
    foot = futil.read_footer(path)
    assert foot["stripes"][0]["nodes"][0][0]["segs"][0]["mode"] > 0
    with oracle.OracleTable(path) as t:
        parts, _ = t.scan_agg([], [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_COL, 0)])
        mask = nulls == 0
        assert parts[0].i128 == int(a[mask].sum())
        assert parts[1].count == int(mask.sum())


def test_canonical_orderkey_boundary(tmp_path):
    """A sorted key crossing a 2^16 boundary inside one chunk needs L=3 —
    the per-chunk XOR-based L choice must cover it exactly."""
    a = np.arange(60000, 70001, dtype=np.int64)    # crosses 65536
    path = str(tmp_path / "ok.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_LZ4)
    foot = futil.read_footer(path)
    first = foot["stripes"][0]["nodes"][0][0]
    assert first["segs"][0]["mode"] == futil.SEGMODE_P_BASE + 3  # crosses 2^16
    with oracle.OracleTable(path) as t:
        parts, _ = t.scan_agg([(0, ca.PRED_GE, 65530)], [(ca.AGG_COUNT_STAR, -1),
                                                         (ca.AGG_SUM_I64, 0)])
        assert parts[0].count == 70001 - 65530
        assert parts[1].i128 == int(np.arange(65530, 70001, dtype=np.int64).sum())


def test_pglz_roundtrip(tmp_path):
    """COMPRESSION_PG_LZ chunks (reference-migrated tables): writer emits the
    reference's ColumnarCompressHeader + pglz stream layout
    (columnar_compression.c:122-151), oracle decodes it back exactly
    (:230-262). Host-side codec by design."""
    n = 30000
    a = (np.arange(n, dtype=np.int64) * 13) % 997      # compressible
    f = np.repeat(np.arange(n // 100, dtype=np.int64), 100)  # runs
    path = str(tmp_path / "pglz.cs")
    ca.write_table(path, [("a", ca.I64, 0), ("f", ca.I64, 0)], [a, f],
                   compression=ca.COMP_PGLZ, chunk_group_row_limit=4000)
    foot = futil.read_footer(path)
    node = foot["stripes"][0]["nodes"][0][0]
    assert node["comp_type"] == ca.COMP_PGLZ
    assert node["value_len"] < node["decompressed_size"]
    with oracle.OracleTable(path) as t:
        parts, _ = t.scan_agg([(0, ca.PRED_GT, 100)],
                              [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1),
                               (ca.AGG_SUM_I64, 1)])
        mask = a > 100
        assert parts[0].i128 == int(a[mask].sum())
        assert parts[1].count == int(mask.sum())
        assert parts[2].i128 == int(f[mask].sum())


def test_pglz_stream_fuzz():
    """pglz decoder vs compressor differential over adversarial byte
    patterns (RLE runs with overlapping matches, random, periodic)."""
    import ctypes
    lib = ctypes.CDLL(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "citus_amd", "libcstripe.so"))
    rng = np.random.default_rng(4242)
    for trial in range(30):
        kind = trial % 4
        n = int(rng.integers(16, 40000))
        if kind == 0:
            data = np.zeros(n, dtype=np.uint8)
        elif kind == 1:
            data = rng.integers(0, 256, n).astype(np.uint8)
        elif kind == 2:
            data = np.tile(np.array([1, 2, 3], dtype=np.uint8), n // 3 + 1)[:n]
        else:
            data = rng.integers(0, 4, n).astype(np.uint8)
        # roundtrip through a tiny tmp table (I8 column, pglz)
        import tempfile
        with tempfile.TemporaryDirectory() as td:
            pth = os.path.join(td, "t.cs")
            ca.write_table(pth, [("b", ca.I8, 0)], [data.view(np.int8)],
                           compression=ca.COMP_PGLZ)
            with oracle.OracleTable(pth) as t:
                v, e = read_all(t, 0, n, np.int8, 10000, stripe_rows=150000)
                np.testing.assert_array_equal(v.view(np.uint8), data)


def test_zstd_restricted_system_decodable(tmp_path):
    """The writer's zstd chunks are RESTRICTED frames (raw literals +
    predefined-FSE sequences, zstd_r.h) tagged CSF_SEGMODE_ZR for the GPU
    lane decoder — and they are STANDARD zstd: the system ZSTD_decompress
    (the call the reference's DecompressBuffer makes,
    columnar_compression.c:207) must reproduce the raw stream exactly."""
    n = 25000
    cases = {
        "li": (np.arange(n, dtype=np.int64) * 7) % 5000,
        "runs": np.repeat(np.arange(n // 50, dtype=np.int64), 50),
        "rand": RNG.integers(-2**62, 2**62, n).astype(np.int64),
    }
    zstd = C.CDLL("libzstd.so.1")
    zstd.ZSTD_decompress.restype = C.c_size_t
    zstd.ZSTD_decompress.argtypes = [C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    zstd.ZSTD_isError.restype = C.c_uint
    zstd.ZSTD_isError.argtypes = [C.c_size_t]
    for name, a in cases.items():
        path = str(tmp_path / f"zr_{name}.cs")
        ca.write_table(path, [("a", ca.I64, 0)], [np.ascontiguousarray(a)],
                       compression=ca.COMP_ZSTD)
        foot = futil.read_footer(path)
        node = foot["stripes"][0]["nodes"][0][0]
        if node["comp_type"] != ca.COMP_ZSTD:
            assert name == "rand"      # incompressible -> raw NONE (ref rule)
            continue
        # P-eligible columns now get a single-segment CANONICAL frame
        # (0x60|L / 0x70); anything else carries generic ZR frames — every
        # one of them must still be plain zstd to the system decoder below
        assert all(futil.is_zr_mode(s["mode"]) for s in node["segs"]), name
        comp = futil.chunk_stream(path, node)
        out = bytearray(node["decompressed_size"])
        pos = 0
        for seg in node["segs"]:
            frame = comp[seg["comp_off"]:seg["comp_off"] + seg["comp_len"]]
            buf = C.create_string_buffer(seg["decomp_len"])
            r = zstd.ZSTD_decompress(buf, seg["decomp_len"], bytes(frame), len(frame))
            assert not zstd.ZSTD_isError(r) and r == seg["decomp_len"], \
                f"{name}: libzstd rejected the restricted frame"
            out[seg["decomp_off"]:seg["decomp_off"] + seg["decomp_len"]] = buf.raw
            pos += seg["decomp_len"]
        nrows0 = node["decompressed_size"] // 8   # first chunk only
        assert bytes(out) == a[:nrows0].tobytes(), name


def test_text_varlena_layout(tmp_path):
    """TEXT columns store EXACTLY the reference's short-varlena serialization
    (SerializeSingleDatum: 1-byte header (total<<1)|1, payload, zero pad to
    the 4-byte att_align_nominal boundary) — byte-level pin on the stream."""
    vals = ["A", "N", "R", "xy", "abc", "Z"]
    slots = ca.text_slots(vals)
    path = str(tmp_path / "t.cs")
    ca.write_table(path, [("f", ca.TEXT, 0)], [slots], compression=ca.COMP_NONE)
    foot = futil.read_footer(path)
    node = foot["stripes"][0]["nodes"][0][0]
    stream = futil.chunk_stream(path, node)
    expect = b""
    for v in vals:
        b = v.encode()
        expect += bytes([((len(b) + 1) << 1) | 1]) + b + b"\0" * (3 - len(b))
    assert stream == expect
    # decode helpers round-trip
    assert [ca.slot_text(sl) for sl in slots] == vals


def test_text_group_and_pred(tmp_path):
    n = 9000
    flags = np.array([["A", "N", "R"][i % 3] for i in range(n)])
    slots = ca.text_slots(flags)
    v = np.arange(n, dtype=np.int64)
    nulls = (np.arange(n) % 11 == 0).astype(np.uint8)
    path = str(tmp_path / "g.cs")
    ca.write_table(path, [("f", ca.TEXT, 0), ("v", ca.I64, 0)], [slots, v],
                   nulls=[nulls, None], compression=ca.COMP_LZ4,
                   chunk_group_row_limit=2000)
    with oracle.OracleTable(path) as t:
        # group by the char(1) column; NULL flags form their own group
        res, _ = t.scan_agg([], [(ca.AGG_SUM_I64, 1), (ca.AGG_COUNT_STAR, -1)],
                            group_cols=(0,))
        for ch in "ANR":
            mask = (flags == ch) & (nulls == 0)
            key = (ord(ch), 0)
            assert res[key][0].i128 == int(v[mask].sum())
            assert res[key][1].count == int(mask.sum())
        nmask = nulls == 1
        assert res[(None, 0)][1].count == int(nmask.sum())
        # predicate: whole-slot equality (f = 'R')
        parts, _ = t.scan_agg([(0, ca.PRED_EQ, ca.text_slot("R"))],
                              [(ca.AGG_COUNT_STAR, -1)])
        assert parts[0].count == int(((flags == "R") & (nulls == 0)).sum())


def test_text_bad_slot_rejected(tmp_path):
    import numpy as np
    bad = np.array([0x00000004], dtype=np.uint32)   # even header: not varlena
    with pytest.raises(ca.CStripeError, match="short varlena"):
        ca.write_table(str(tmp_path / "b.cs"), [("f", ca.TEXT, 0)], [bad])


def test_zstd_large_segment_frames(tmp_path):
    """restricted-zstd frames above the 2-byte frame-content-size limit
    (65 791 B) must use the 4-byte FCS form — large zstd segment targets
    round-trip through libzstd exactly."""
    n = 40_000                       # 320 KB chunk at 256 KB segment target
    a = (np.arange(n, dtype=np.int64) * 17) % 911
    path = str(tmp_path / "bigseg.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_ZSTD,
                   lz4_seg_target_kb=256, chunk_group_row_limit=40_000,
                   stripe_row_limit=40_000)
    foot = futil.read_footer(path)
    node = foot["stripes"][0]["nodes"][0][0]
    assert node["comp_type"] == ca.COMP_ZSTD
    assert any(s2["decomp_len"] > 65_791 for s2 in node["segs"])
    with oracle.OracleTable(path) as t:     # oracle decodes via libzstd
        parts, _ = t.scan_agg([], [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1)])
        assert parts[0].i128 == int(a.sum())
        assert parts[1].count == n


def test_zstd_canonical_layout(tmp_path):
    """Canonical restricted-zstd frames (zr_canon_p / zr_canon_const): value
    bytes sit at the CLOSED-FORM literal-section positions format.h documents
    (csf_canon_zrp_pos: value 0 at 15, value j>=1 low-L at 23+(j-1)*L), one
    segment, mode 0x60|L / 0x70 — and every frame is still plain zstd to the
    system ZSTD_decompress (the reference's DecompressBuffer call,
    columnar_compression.c:207)."""
    n = 25000
    zstd = C.CDLL("libzstd.so.1")
    zstd.ZSTD_decompress.restype = C.c_size_t
    zstd.ZSTD_decompress.argtypes = [C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    zstd.ZSTD_isError.restype = C.c_uint
    zstd.ZSTD_isError.argtypes = [C.c_size_t]
    cases = [
        ("p2", (np.arange(n, dtype=np.int64) * 7) % 5000, 0x62, 2),
        ("p1", np.asarray((np.arange(n) * 31) % 200, dtype=np.int64), 0x61, 1),
        ("const", np.full(n, 0x1234, dtype=np.int64), 0x70, 0),
    ]
    for name, a, want_mode, L in cases:
        path = str(tmp_path / f"zc_{name}.cs")
        ca.write_table(path, [("a", ca.I64, 0)], [np.ascontiguousarray(a)],
                       compression=ca.COMP_ZSTD)
        foot = futil.read_footer(path)
        node = foot["stripes"][0]["nodes"][0][0]
        assert node["comp_type"] == ca.COMP_ZSTD, name
        assert node["n_segs"] == 1 and node["segs"][0]["mode"] == want_mode, name
        comp = futil.chunk_stream(path, node)
        rows0 = node["decompressed_size"] // 8
        # closed-form position pin against the raw stream bytes
        assert comp[15:23] == a[:1].tobytes(), name
        if L:
            for j in (1, 2, 777, rows0 - 1):
                pos = 23 + (j - 1) * L
                assert comp[pos:pos + L] == a[j:j + 1].tobytes()[:L], (name, j)
        # libzstd reproduces the raw stream from the canonical frame
        buf = C.create_string_buffer(node["decompressed_size"])
        r = zstd.ZSTD_decompress(buf, node["decompressed_size"],
                                 bytes(comp), len(comp))
        assert not zstd.ZSTD_isError(r) and r == node["decompressed_size"], name
        assert buf.raw == a[:rows0].tobytes(), name
        # and the oracle reads the same values end to end
        with oracle.OracleTable(path) as t:
            v, e = read_all(t, 0, n, np.int64, 10000, stripe_rows=150000)
            np.testing.assert_array_equal(v, a)


def test_zstd_canonical_width4_flags(tmp_path):
    """Width-4 canonical zstd (zr_canon_b4): char(1) short-varlena slots
    [0x05 ch 00 00] vary only in byte 1 — per row ONE literal byte at the
    closed-form position csf_canon_zr4b_pos documents (15 + (j==0 ? k :
    j+3+k)), mode 0x80|k; all-equal slots -> 0x84. Both still plain zstd
    to ZSTD_decompress. (LZ4's 4-byte minimum match cannot express this
    parse — zstd's is 3.)"""
    n = 25000
    zstd = C.CDLL("libzstd.so.1")
    zstd.ZSTD_decompress.restype = C.c_size_t
    zstd.ZSTD_decompress.argtypes = [C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    zstd.ZSTD_isError.restype = C.c_uint
    zstd.ZSTD_isError.argtypes = [C.c_size_t]
    flags = np.array([ord("ANR"[i % 3]) for i in range(n)], dtype=np.uint32)
    slots_var = (0x05 | (flags << 8)).astype(np.uint32)        # byte 1 varies
    slots_const = np.full(n, 0x05 | (ord("X") << 8), dtype=np.uint32)
    lowbyte = np.arange(n, dtype=np.uint32) % 200              # byte 0 varies
    for name, slots, want_mode, k in [("flags", slots_var, 0x81, 1),
                                      ("const", slots_const, 0x84, None),
                                      ("i32low", lowbyte, 0x80, 0)]:
        typ = ca.TEXT if name != "i32low" else ca.I32
        col = slots.view(np.int32)
        path = str(tmp_path / f"z4_{name}.cs")
        ca.write_table(path, [("f", typ, 0)], [np.ascontiguousarray(col)],
                       compression=ca.COMP_ZSTD)
        foot = futil.read_footer(path)
        node = foot["stripes"][0]["nodes"][0][0]
        assert node["comp_type"] == ca.COMP_ZSTD, name
        assert node["n_segs"] == 1 and node["segs"][0]["mode"] == want_mode, \
            (name, hex(node["segs"][0]["mode"]))
        comp = futil.chunk_stream(path, node)
        rows0 = node["decompressed_size"] // 4
        raw0 = slots[:rows0].tobytes()
        # closed-form position pin
        assert comp[15:19] == raw0[:4], name
        if k is not None:
            for j in (0, 1, 2, 777, rows0 - 1):
                pos = 15 + (k if j == 0 else j + 3 + k)
                assert comp[pos] == (slots[j] >> (8 * k)) & 0xFF, (name, j)
        # system-zstd round trip
        buf = C.create_string_buffer(node["decompressed_size"])
        r = zstd.ZSTD_decompress(buf, node["decompressed_size"],
                                 bytes(comp), len(comp))
        assert not zstd.ZSTD_isError(r) and r == node["decompressed_size"], name
        assert buf.raw == raw0, name
        # oracle end-to-end
        with oracle.OracleTable(path) as t:
            v, e = read_all(t, 0, n, np.int32, 10000, stripe_rows=150000)
            np.testing.assert_array_equal(v.view(np.uint32), slots)


def test_zstd_canonical_boundary_sizes(tmp_path):
    """Canonical-zstd emitters across size boundaries (nseq 1-byte/2-byte
    forms, minimum n, chunk tails): every emitted frame must reproduce the
    raw stream through the system libzstd, and the oracle must read the
    values back exactly."""
    zstd = C.CDLL("libzstd.so.1")
    zstd.ZSTD_decompress.restype = C.c_size_t
    zstd.ZSTD_decompress.argtypes = [C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    zstd.ZSTD_isError.restype = C.c_uint
    zstd.ZSTD_isError.argtypes = [C.c_size_t]
    rng = np.random.default_rng(11)
    sizes = [3, 4, 5, 127, 128, 129, 130, 131, 999, 10000, 10003]
    for n in sizes:
        for case in ("p1", "p4", "c8", "b4", "c4"):
            if case in ("p1", "p4", "c8"):
                if case == "p1":
                    a = rng.integers(0, 200, n).astype(np.int64)
                elif case == "p4":
                    a = rng.integers(0, 2**31, n).astype(np.int64)
                else:
                    a = np.full(n, -12345, dtype=np.int64)
                defs, cols, dt, w = [("a", ca.I64, 0)], [a], np.int64, 8
            else:
                if case == "b4":
                    ch = rng.integers(65, 68, n).astype(np.uint32)
                    a = (0x05 | (ch << 8)).view(np.int32)
                else:
                    a = np.full(n, 0x05 | (70 << 8), dtype=np.uint32).view(np.int32)
                defs, cols, dt, w = [("a", ca.TEXT, 0)], [a], np.int32, 4
            path = str(tmp_path / f"b_{case}_{n}.cs")
            ca.write_table(path, defs, cols, compression=ca.COMP_ZSTD,
                           chunk_group_row_limit=10000)
            foot = futil.read_footer(path)
            raw = np.asarray(cols[0])
            pos = 0
            for st in foot["stripes"]:
                for ci, nd in enumerate(st["nodes"][0]):
                    rows = nd["decompressed_size"] // w
                    if nd["comp_type"] == ca.COMP_ZSTD:
                        comp = futil.chunk_stream(path, nd)
                        out = bytearray(nd["decompressed_size"])
                        for sg in nd["segs"]:
                            fr = comp[sg["comp_off"]:sg["comp_off"] + sg["comp_len"]]
                            buf = C.create_string_buffer(sg["decomp_len"])
                            r = zstd.ZSTD_decompress(buf, sg["decomp_len"],
                                                     bytes(fr), len(fr))
                            assert not zstd.ZSTD_isError(r) and r == sg["decomp_len"], \
                                (case, n, hex(sg["mode"]))
                            out[sg["decomp_off"]:sg["decomp_off"] + sg["decomp_len"]] = buf.raw
                        assert bytes(out) == raw[pos:pos + rows].tobytes(), (case, n)
                    pos += rows
            with oracle.OracleTable(path) as t:
                v, e = read_all(t, 0, n, dt, 10000, stripe_rows=150000)
                np.testing.assert_array_equal(v, raw[:n].astype(dt) if dt != np.int32 else raw[:n])


def test_zstd_canonical_random_fuzz(tmp_path):
    """Seeded random sweep over canonical-zstd eligibility shapes: random
    L in 1..4 (width-8), varying-byte k in 0..3 (width-4), random sizes,
    optional NULLs — every zstd chunk the writer emits must reproduce its
    raw stream through libzstd, and the oracle read must be exact."""
    zstd = C.CDLL("libzstd.so.1")
    zstd.ZSTD_decompress.restype = C.c_size_t
    zstd.ZSTD_decompress.argtypes = [C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    zstd.ZSTD_isError.restype = C.c_uint
    zstd.ZSTD_isError.argtypes = [C.c_size_t]
    rng = np.random.default_rng(99)
    for trial in range(60):
        n = int(rng.integers(3, 30000))
        wide = rng.random() < 0.5
        with_nulls = rng.random() < 0.3
        if wide:
            L = int(rng.integers(1, 5))
            base = int(rng.integers(-2**40, 2**40)) & ~((1 << (8 * L)) - 1)
            a = (base | rng.integers(0, 1 << (8 * L), n, dtype=np.int64)).astype(np.int64)
            defs, dt, w = [("a", ca.I64, 0)], np.int64, 8
        else:
            k = int(rng.integers(0, 4))
            base = int(rng.integers(0, 2**32)) & ~(0xFF << (8 * k))
            a = (base | (rng.integers(0, 256, n, dtype=np.uint32) << (8 * k))) \
                .astype(np.uint32).view(np.int32)
            defs, dt, w = [("a", ca.I32, 0)], np.int32, 4
        nulls = [(rng.random(n) < 0.2).astype(np.uint8)] if with_nulls else None
        path = str(tmp_path / f"fz{trial}.cs")
        ca.write_table(path, defs, [np.ascontiguousarray(a)], nulls=nulls,
                       compression=ca.COMP_ZSTD,
                       chunk_group_row_limit=int(rng.choice([1000, 3000, 10000])))
        foot = futil.read_footer(path)
        for st in foot["stripes"]:
            for nd in st["nodes"][0]:
                if nd["comp_type"] != ca.COMP_ZSTD:
                    continue
                comp = futil.chunk_stream(path, nd)
                for sg in nd["segs"]:
                    fr = comp[sg["comp_off"]:sg["comp_off"] + sg["comp_len"]]
                    buf = C.create_string_buffer(sg["decomp_len"])
                    r = zstd.ZSTD_decompress(buf, sg["decomp_len"],
                                             bytes(fr), len(fr))
                    assert not zstd.ZSTD_isError(r) and r == sg["decomp_len"], \
                        (trial, hex(sg["mode"]))
        with oracle.OracleTable(path) as t:
            v, e = read_all(t, 0, n, dt, foot["chunk_row_limit"],
                            stripe_rows=150000)
            if nulls is None:
                np.testing.assert_array_equal(v, a)
            else:
                keep = nulls[0] == 0
                np.testing.assert_array_equal(v[keep], a[keep])
                np.testing.assert_array_equal(e, 1 - nulls[0])
