"""
citus_amd — MI355X-native implementation of Citus's columnar scan +
partial-aggregate hot path (chunk-group read -> LZ4 decode -> qual filter ->
partial aggregate -> combine), behind the reference's scan/combine surfaces
restated as a C ABI (include/cstripe.h).

This package is ctypes plumbing over libcstripe.so (hand-written HIP/CDNA4
kernels + host C++). The GPU path requires a visible MI355X and fails loudly
without one; the CPU reference restatement lives in oracle/ and is test
infrastructure only.
"""
import ctypes as C
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libcstripe.so")

if not os.path.exists(_LIB_PATH):
    raise ImportError(
        "libcstripe.so not built — run `make -C citus_amd/csrc` "
        "(or __graft_entry__.build())")

# Two HIP runtime instances in one process (torch's bundled copy + the
# system /opt/rocm one) break device enumeration for whichever initializes
# second — in either direction on some boxes. Both ship the same soname
# (libamdhip64.so.7), so preloading torch's copy globally makes
# libcstripe's NEEDED entry resolve to the SAME mapped runtime: one HIP
# instance shared by both stacks. Plain-C ABI consumers (no torch in the
# process) are unaffected and keep the system runtime.
try:
    import torch as _torch
    _torch_hip = os.path.join(os.path.dirname(_torch.__file__), "lib",
                              "libamdhip64.so")
    if os.path.exists(_torch_hip):
        C.CDLL(_torch_hip, mode=C.RTLD_GLOBAL)
except Exception:
    pass

_lib = C.CDLL(_LIB_PATH)

# ---- enums (include/cstripe.h) ----
I8, I16, I32, I64, F32, F64, TEXT = 1, 2, 3, 4, 5, 6, 7
COMP_NONE, COMP_PGLZ, COMP_LZ4, COMP_ZSTD = 0, 1, 2, 3
PRED_LT, PRED_LE, PRED_GT, PRED_GE, PRED_EQ, PRED_NE = range(6)
(AGG_COUNT_STAR, AGG_COUNT_COL, AGG_SUM_I64, AGG_SUM_F64,
 AGG_MIN_I64, AGG_MAX_I64, AGG_MIN_F64, AGG_MAX_F64,
 AGG_SUM_PROD_I64, AGG_SUM_DISC_I64, AGG_SUM_DISC_TAX_I64) = range(11)

OK, ERR, ERR_IO, ERR_FORMAT, ERR_NOGPU, ERR_ARG, END = 0, -1, -2, -3, -4, -5, 1


class ColDef(C.Structure):
    _fields_ = [("name", C.c_char * 32), ("type", C.c_uint8),
                ("scale", C.c_uint8), ("_pad", C.c_uint8 * 6)]


class Options(C.Structure):
    _fields_ = [("stripe_row_limit", C.c_uint64),
                ("chunk_group_row_limit", C.c_uint32),
                ("compression", C.c_uint8), ("compression_level", C.c_int8),
                ("lz4_seg_target_kb", C.c_uint16),
                ("lz4_seg_target_bytes", C.c_uint32),
                ("lz4_min_match", C.c_uint8), ("canonical", C.c_uint8),
                ("_pad2", C.c_uint8 * 6)]


class Pred(C.Structure):
    _fields_ = [("column", C.c_uint32), ("op", C.c_uint32),
                ("ival", C.c_int64), ("fval", C.c_double),
                ("or_group", C.c_uint32), ("_pad", C.c_uint32)]


class AggSpec(C.Structure):
    _fields_ = [("kind", C.c_uint32), ("col_a", C.c_int32), ("col_b", C.c_int32),
                ("col_c", C.c_int32), ("one", C.c_int64)]


class Partial(C.Structure):
    _fields_ = [("i128_lo", C.c_int64), ("i128_hi", C.c_int64), ("f64", C.c_double),
                ("count", C.c_int64), ("is_null", C.c_uint8), ("_pad", C.c_uint8 * 7)]

    @property
    def i128(self):
        """signed 128-bit view (sums; min/max are sign-extended into hi)"""
        return (self.i128_hi << 64) | (self.i128_lo & ((1 << 64) - 1))

    def as_dict(self):
        return {"i128": self.i128, "f64": self.f64, "count": self.count,
                "is_null": bool(self.is_null)}


class Batch(C.Structure):
    _fields_ = [("n_rows", C.c_uint32), ("first_row_number", C.c_uint64),
                ("col_values", C.POINTER(C.c_void_p)),
                ("col_nulls", C.POINTER(C.POINTER(C.c_uint8)))]


class GroupResult(C.Structure):
    _fields_ = [("n_groups", C.c_uint32), ("keys", C.c_uint32 * 64)]


GROUP_KEY_NULL = 0x100          # 9-bit key encoding: bit 8 = NULL key


def decode_group_key(kv):
    """(k0, k1) from the packed 9-bit-per-column encoding; NULL keys (their
    own group, like the reference HashAggregate) decode to None."""
    e0 = kv & 0x1FF
    e1 = (kv >> 9) & 0x1FF
    return (None if e0 & GROUP_KEY_NULL else e0,
            None if e1 & GROUP_KEY_NULL else e1)


def encode_group_key(k):
    e0 = GROUP_KEY_NULL if k[0] is None else k[0]
    k1 = k[1] if len(k) > 1 else 0
    e1 = GROUP_KEY_NULL if k1 is None else k1
    return e0 | (e1 << 9)


def _sig(name, res, args):
    f = getattr(_lib, name)
    f.restype = res
    f.argtypes = args
    return f


_errmsg = _sig("cstripe_errmsg", C.c_char_p, [])
_default_options = _sig("cstripe_default_options", None, [C.POINTER(Options)])
_write_begin = _sig("cstripe_write_begin", C.c_void_p,
                    [C.c_char_p, C.POINTER(ColDef), C.c_uint32, C.POINTER(Options)])
_write_rows = _sig("cstripe_write_rows", C.c_int,
                   [C.c_void_p, C.c_uint64, C.POINTER(C.c_void_p), C.POINTER(C.c_void_p)])
_write_end = _sig("cstripe_write_end", C.c_int, [C.c_void_p])
_write_rows_device = _sig("cstripe_write_rows_device", C.c_int,
                          [C.c_void_p, C.c_uint64, C.POINTER(C.c_void_p)])
_open = _sig("cstripe_open", C.c_void_p, [C.c_char_p])
_close = _sig("cstripe_close", None, [C.c_void_p])
_row_count = _sig("cstripe_row_count", C.c_uint64, [C.c_void_p])
_column_count = _sig("cstripe_column_count", C.c_uint32, [C.c_void_p])
_stripe_count = _sig("cstripe_stripe_count", C.c_uint32, [C.c_void_p])
_column_def = _sig("cstripe_column_def", C.c_int, [C.c_void_p, C.c_uint32, C.POINTER(ColDef)])
_scan_begin = _sig("cstripe_scan_begin", C.c_void_p,
                   [C.c_void_p, C.c_uint64, C.POINTER(Pred), C.c_uint32])
_scan_end = _sig("cstripe_scan_end", None, [C.c_void_p])
_filtered = _sig("cstripe_scan_chunk_groups_filtered", C.c_int64, [C.c_void_p])
_gpu_stage = _sig("cstripe_gpu_stage", C.c_int, [C.c_void_p, C.c_int])
_staged_bytes = _sig("cstripe_gpu_staged_bytes", C.c_uint64, [C.c_void_p])
_scan_agg = _sig("cstripe_scan_agg", C.c_int,
                 [C.c_void_p, C.POINTER(AggSpec), C.c_uint32, C.POINTER(Partial)])
_scan_agg_grouped = _sig("cstripe_scan_agg_grouped", C.c_int,
                         [C.c_void_p, C.POINTER(AggSpec), C.c_uint32,
                          C.POINTER(C.c_uint32), C.c_uint32, C.POINTER(GroupResult),
                          C.POINTER(Partial)])
_next_batch = _sig("cstripe_scan_next_batch", C.c_int, [C.c_void_p, C.POINTER(Batch)])
_read_row = _sig("cstripe_read_row", C.c_int,
                 [C.c_void_p, C.c_uint64, C.POINTER(C.c_void_p),
                  C.POINTER(C.c_uint8)])
_rewind = _sig("cstripe_scan_rewind", C.c_int, [C.c_void_p])
_last_fused = _sig("cstripe_scan_last_fused", C.c_int, [C.c_void_p])
_last_kernel_ms = _sig("cstripe_scan_last_kernel_ms", C.c_double, [C.c_void_p])
_last_decode_ms = _sig("cstripe_scan_last_decode_kernel_ms", C.c_double, [C.c_void_p])
_last_agg_ms = _sig("cstripe_scan_last_agg_kernel_ms", C.c_double, [C.c_void_p])
_combine = _sig("cagg_combine", C.c_int,
                [C.POINTER(AggSpec), C.c_uint32, C.POINTER(Partial), C.c_uint32,
                 C.POINTER(Partial)])
_gpu_available = _sig("cstripe_gpu_available", C.c_int, [])
_comm_uid = _sig("cagg_comm_unique_id", C.c_int, [C.c_uint8 * 128])
_comm_init = _sig("cagg_comm_init", C.c_int,
                  [C.POINTER(C.c_void_p), C.c_int, C.c_int, C.c_uint8 * 128, C.c_int])
_comm_destroy = _sig("cagg_comm_destroy", None, [C.c_void_p])
_comm_allgather = _sig("cagg_allgather", C.c_int,
                       [C.c_void_p, C.c_void_p, C.c_uint64, C.c_void_p])
_combine_rccl = _sig("cagg_combine_rccl", C.c_int,
                     [C.c_void_p, C.POINTER(AggSpec), C.c_uint32,
                      C.POINTER(Partial), C.POINTER(Partial)])
_gen_lineitem = _sig("csbench_gen_lineitem", C.c_int,
                     [C.c_char_p, C.c_uint64, C.c_uint64, C.c_int, C.c_int, C.c_int,
                      C.c_uint64, C.c_uint32])
_gen_lineitem2 = _sig("csbench_gen_lineitem2", C.c_int,
                      [C.c_char_p, C.c_uint64, C.c_uint64, C.c_int, C.c_int, C.c_int,
                       C.c_uint64, C.c_uint32, C.c_int, C.c_int])
_gen_shards = _sig("csbench_gen_lineitem_shards", C.c_int,
                   [C.c_char_p, C.c_uint64, C.c_uint32, C.c_uint64, C.c_int, C.c_int,
                    C.c_int, C.c_uint64, C.c_uint32, C.c_int, C.c_int, C.c_int])
_expected_q6 = _sig("csbench_expected_q6", C.c_int,
                    [C.c_uint64, C.c_uint64, C.POINTER(C.c_int64),
                     C.POINTER(C.c_int64), C.POINTER(C.c_int64)])
_expected_q1 = _sig("csbench_expected_q1", C.c_int,
                    [C.c_uint64, C.c_uint64, C.POINTER(C.c_int64),
                     C.POINTER(C.c_int64), C.POINTER(C.c_int64)])


def errmsg():
    return _errmsg().decode()


def gpu_available():
    return bool(_gpu_available())


class CStripeError(RuntimeError):
    pass


def _check(rc, what):
    if rc != OK:
        raise CStripeError(f"{what} failed (rc={rc}): {errmsg()}")


def default_options(**kw):
    o = Options()
    _default_options(C.byref(o))
    if "lz4_seg_target_kb" in kw and "lz4_seg_target_bytes" not in kw:
        kw["lz4_seg_target_bytes"] = 0   # the coarse knob overrides the default
    for k, v in kw.items():
        setattr(o, k, v)
    return o


def text_slot(sval):
    """4-byte short-varlena slot (the reference's stored datum bytes) for a
    text value of <= 3 payload bytes: hdr = ((len+1)<<1)|1, payload, zero pad.
    Returns the slot as a little-endian u32 (use for TEXT predicates)."""
    b = sval.encode() if isinstance(sval, str) else bytes(sval)
    assert 1 <= len(b) <= 3
    slot = bytes([((len(b) + 1) << 1) | 1]) + b + b"\0" * (3 - len(b))
    return int.from_bytes(slot, "little")


def text_slots(values):
    """numpy uint32 array of varlena slots for a sequence of short strings"""
    import numpy as np
    return np.array([text_slot(v) for v in values], dtype=np.uint32)


def slot_text(slot):
    """decode a u32 varlena slot back to its string"""
    b = int(slot).to_bytes(4, "little")
    n = (b[0] >> 1) - 1
    return b[1:1 + n].decode()


def make_coldefs(defs):
    """defs: list of (name, type, scale)"""
    arr = (ColDef * len(defs))()
    for i, (name, typ, scale) in enumerate(defs):
        arr[i].name = name.encode()
        arr[i].type = typ
        arr[i].scale = scale
    return arr


def write_table(path, defs, columns, nulls=None, **opt_kw):
    """Write numpy columns to a stripe file. columns: list of np arrays
    (dtype matching coldef types); nulls: optional list of uint8 arrays."""
    import numpy as np
    opts = default_options(**opt_kw)
    cols = make_coldefs(defs)
    w = _write_begin(path.encode(), cols, len(defs), C.byref(opts))
    if not w:
        raise CStripeError("write_begin: " + errmsg())
    n = len(columns[0])
    vals = (C.c_void_p * len(defs))()
    for i, a in enumerate(columns):
        a = np.ascontiguousarray(a)
        columns[i] = a
        vals[i] = a.ctypes.data_as(C.c_void_p).value
    nl = None
    if nulls is not None:
        nl = (C.c_void_p * len(defs))()
        for i, a in enumerate(nulls):
            nl[i] = a.ctypes.data_as(C.c_void_p).value if a is not None else None
    _check(_write_rows(w, n, vals, nl), "write_rows")
    _check(_write_end(w), "write_end")


def write_table_device(path, defs, dev_ptrs, n_rows, **opt_kw):
    """Device-side write (cstripe_write_rows_device): dev_ptrs are raw HBM
    addresses of column arrays (e.g. torch tensor.data_ptr()); full chunks
    compress on the GPU, only compressed bytes cross PCIe."""
    opts = default_options(**opt_kw)
    cols = make_coldefs(defs)
    w = _write_begin(path.encode(), cols, len(defs), C.byref(opts))
    if not w:
        raise CStripeError("write_begin: " + errmsg())
    vals = (C.c_void_p * len(defs))(*[int(p) for p in dev_ptrs])
    rc = _write_rows_device(w, n_rows, vals)
    if rc != OK:
        _write_end(w)
        raise CStripeError(f"write_rows_device failed (rc={rc}): {errmsg()}")
    _check(_write_end(w), "write_end")


def gen_lineitem_shards(dirpath, total_rows, n_shards, base_seed=42,
                        compression=COMP_LZ4, level=3, seg_kb=0, stripe_rows=0,
                        chunk_rows=0, min_match=0, canonical=1, threads=0):
    """Parallel sharded generation: n_shards files shardNN.cs under dirpath
    (the config-3 shard-directory shape), seeds base_seed + i."""
    os.makedirs(dirpath, exist_ok=True)
    _check(_gen_shards(dirpath.encode(), total_rows, n_shards, base_seed,
                       compression, level, seg_kb, stripe_rows, chunk_rows,
                       min_match, canonical, threads), "gen_lineitem_shards")


def expected_q6(n_rows, seed=42):
    """Exact Q6 (revenue int128 at scale 4, count) recomputed from the
    generator streams on host CPU — the bench's in-run parity pin."""
    lo, hi, cnt = C.c_int64(), C.c_int64(), C.c_int64()
    _check(_expected_q6(n_rows, seed, C.byref(lo), C.byref(hi), C.byref(cnt)),
           "expected_q6")
    return (hi.value << 64) | (lo.value & ((1 << 64) - 1)), cnt.value


def expected_q1(n_rows, seed=42):
    """Exact Q1 per-(returnflag, linestatus) sums/counts from the generator.
    Returns {(rf, ls): [sum_qty, sum_price, sum_disc_price, sum_charge, count]}."""
    lo = (C.c_int64 * 24)()
    hi = (C.c_int64 * 24)()
    cnt = (C.c_int64 * 6)()
    _check(_expected_q1(n_rows, seed, lo, hi, cnt), "expected_q1")
    out = {}
    for rf in range(3):
        for ls in range(2):
            g = rf * 2 + ls
            sums = [(hi[g * 4 + k] << 64) | (lo[g * 4 + k] & ((1 << 64) - 1))
                    for k in range(4)]
            out[(rf, ls)] = sums + [cnt[g]]
    return out


def gen_lineitem(path, n_rows, seed=42, compression=COMP_LZ4, level=3, seg_kb=0,
                 stripe_rows=0, chunk_rows=0, seg_bytes=0, min_match=0,
                 canonical=1):
    if seg_bytes:
        seg_kb = -int(seg_bytes)      # negative seg_kb = bytes (csbench_gen_lineitem)
    _check(_gen_lineitem2(path.encode(), n_rows, seed, compression, level, seg_kb,
                          stripe_rows, chunk_rows, min_match, canonical),
           "gen_lineitem")


LINEITEM_COLS = {"l_orderkey": 0, "l_quantity": 1, "l_extendedprice": 2,
                 "l_discount": 3, "l_tax": 4, "l_shipdate": 5,
                 "l_returnflag": 6, "l_linestatus": 7}


class Reader:
    def __init__(self, path):
        self._h = _open(path.encode())
        if not self._h:
            raise CStripeError("open: " + errmsg())
        self.path = path

    def close(self):
        if self._h:
            _close(self._h)
            self._h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    @property
    def row_count(self):
        return _row_count(self._h)

    @property
    def column_count(self):
        return _column_count(self._h)

    @property
    def stripe_count(self):
        return _stripe_count(self._h)

    def column_def(self, i):
        d = ColDef()
        _check(_column_def(self._h, i, C.byref(d)), "column_def")
        return d.name.decode(), d.type, d.scale

    def scan(self, cols_mask=0, preds=()):
        return Scan(self, cols_mask, preds)


def make_preds(preds):
    """preds: (column, op, value) or (column, op, value, or_group) tuples.
    A nonzero or_group groups predicates into a disjunction (CNF; see
    include/cstripe.h)."""
    arr = (Pred * max(1, len(preds)))()
    for i, p in enumerate(preds):
        arr[i].column = p[0]
        arr[i].op = p[1]
        v = p[2]
        if isinstance(v, float):
            arr[i].fval = v
            arr[i].ival = 0
        else:
            arr[i].ival = int(v)
            arr[i].fval = 0.0
        arr[i].or_group = p[3] if len(p) > 3 else 0
    return arr


def agg_cols_mask(aggs):
    """Projection bitmask covering every column the agg specs read — the
    caller-side analog of ColumnarAttrNeeded (columnar_customscan.c:1813-1851):
    projection is fixed at scan_begin, before aggs are bound."""
    m = 0
    for a in aggs:
        for c in a[1:4]:
            if isinstance(c, int) and c >= 0:
                m |= 1 << c
    return m


def make_aggs(aggs):
    """aggs: list of (kind, col_a[, col_b[, col_c[, one]]])"""
    arr = (AggSpec * len(aggs))()
    for i, a in enumerate(aggs):
        a = tuple(a) + (-1, -1, -1, 0)[len(a) - 1:]
        arr[i].kind, arr[i].col_a, arr[i].col_b, arr[i].col_c, arr[i].one = a[:5]
    return arr


class Scan:
    def __init__(self, reader, cols_mask, preds):
        self._preds = make_preds(preds)
        self._h = _scan_begin(reader._h, cols_mask, self._preds, len(preds))
        if not self._h:
            raise CStripeError("scan_begin: " + errmsg())
        self.reader = reader
        self._reader = reader
        # pred columns are implicitly projected (scan_begin does the same)
        self._mask = cols_mask
        for p in preds:
            self._mask |= 1 << p[0]
        if self._mask == 0:
            self._mask = 1

    def end(self):
        if self._h:
            _scan_end(self._h)
            self._h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.end()

    @property
    def chunk_groups_filtered(self):
        return _filtered(self._h)

    @property
    def staged_bytes(self):
        return _staged_bytes(self._h)

    @property
    def last_fused(self):
        return bool(_last_fused(self._h))

    @property
    def last_kernel_ms(self):
        return _last_kernel_ms(self._h)

    @property
    def last_decode_ms(self):
        return _last_decode_ms(self._h)

    @property
    def last_agg_ms(self):
        return _last_agg_ms(self._h)

    def stage(self, device=-1):
        _check(_gpu_stage(self._h, device), "gpu_stage")

    def agg(self, aggs):
        """aggs: list of (kind, col_a[, col_b[, col_c[, one]]]) -> [Partial]"""
        arr = make_aggs(aggs)
        out = (Partial * len(aggs))()
        _check(_scan_agg(self._h, arr, len(aggs), out), "scan_agg")
        return list(out)

    def agg_grouped(self, aggs, group_cols):
        arr = make_aggs(aggs)
        gc = (C.c_uint32 * len(group_cols))(*group_cols)
        gr = GroupResult()
        out = (Partial * (64 * len(aggs)))()
        _check(_scan_agg_grouped(self._h, arr, len(aggs), gc, len(group_cols),
                                 C.byref(gr), out), "scan_agg_grouped")
        res = {}
        for g in range(gr.n_groups):
            key = decode_group_key(gr.keys[g])
            res[key] = [out[g * len(aggs) + a] for a in range(len(aggs))]
        return res

    def next_batch(self, col_arrays, null_arrays=None):
        """col_arrays: dict col_index -> numpy array (capacity chunk rows).
        Returns (n_rows, first_row_number) or None at end."""
        ncols = self.reader.column_count
        b = Batch()
        vals = (C.c_void_p * ncols)()
        for ci, a in col_arrays.items():
            vals[ci] = a.ctypes.data_as(C.c_void_p).value
        b.col_values = vals
        if null_arrays:
            nl = (C.POINTER(C.c_uint8) * ncols)()
            for ci, a in null_arrays.items():
                nl[ci] = a.ctypes.data_as(C.POINTER(C.c_uint8))
            b.col_nulls = nl
        rc = _next_batch(self._h, C.byref(b))
        if rc == END:
            return None
        _check(rc, "next_batch")
        return b.n_rows, b.first_row_number

    def read_row(self, row_number):
        """Random access (ColumnarReadRowByRowNumber): {col: value-or-None}
        over the projected columns, or None when the row does not exist.
        Requires a predicate-free scan."""
        import numpy as np
        r = self._reader
        ncols = r.column_count
        DT = {1: np.int8, 2: np.int16, 3: np.int32, 4: np.int64,
              5: np.float32, 6: np.float64, 7: np.uint32}
        bufs, vp = {}, (C.c_void_p * ncols)()
        nulls = (C.c_uint8 * ncols)()
        for c in range(ncols):
            if not (self._mask >> c) & 1:
                continue
            a = np.zeros(1, dtype=DT[r.column_def(c)[1]])
            bufs[c] = a
            vp[c] = a.ctypes.data_as(C.c_void_p).value
        rc = _read_row(self._h, row_number, vp, nulls)
        if rc == END:
            return None
        _check(rc, "read_row")
        return {c: (None if nulls[c] else bufs[c][0].item())
                for c in bufs}

    def rewind(self):
        _check(_rewind(self._h), "rewind")


class RcclComm:
    """C-ABI RCCL communicator (cagg_comm): the data-path collective of the
    combine step runs in host C over librccl; the launcher only distributes
    the 128-byte unique id (ncclCommInitRank's own bootstrap contract)."""

    def __init__(self, n_ranks, rank, uid_bytes, device=-1):
        uid = (C.c_uint8 * 128)(*uid_bytes)
        h = C.c_void_p()
        _check(_comm_init(C.byref(h), n_ranks, rank, uid, device), "cagg_comm_init")
        self._h = h
        self.rank = rank
        self.n_ranks = n_ranks

    @staticmethod
    def unique_id():
        uid = (C.c_uint8 * 128)()
        _check(_comm_uid(uid), "cagg_comm_unique_id")
        return bytes(uid)

    def close(self):
        if self._h:
            _comm_destroy(self._h)
            self._h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def allgather_bytes(self, data):
        src = (C.c_uint8 * len(data)).from_buffer_copy(data)
        dst = (C.c_uint8 * (len(data) * self.n_ranks))()
        _check(_comm_allgather(self._h, src, len(data), dst), "cagg_allgather")
        return bytes(dst)

    def combine(self, aggs, parts):
        """cagg_combine_rccl: gather + strict merge, in C, over RCCL."""
        a = make_aggs(aggs)
        local = (Partial * len(parts))(*parts)
        out = (Partial * len(parts))()
        _check(_combine_rccl(self._h, a, len(aggs), local, out),
               "cagg_combine_rccl")
        return list(out)


def combine(aggs, parts_list):
    """parts_list: list (per shard/GPU) of lists of Partial -> [Partial].
    The coordinator-merge step (cagg_combine)."""
    n_aggs = len(aggs)
    arr = make_aggs(aggs)
    flat = (Partial * (n_aggs * len(parts_list)))()
    for p, parts in enumerate(parts_list):
        for a in range(n_aggs):
            flat[p * n_aggs + a] = parts[a]
    out = (Partial * n_aggs)()
    _check(_combine(arr, n_aggs, flat, len(parts_list), out), "combine")
    return list(out)
