"""
oracle — ctypes wrapper over liboracle.so, the CPU restatement of the Citus
columnar read + partial-aggregate path (see coracle.c header for the
function-by-function reference citations and parity pins).

TEST INFRASTRUCTURE ONLY: imported by tests/, __graft_entry__.smoke() and
bench.py's cpu_baseline leg — never by the product path.
"""
import ctypes as C
import os

from citus_amd import (Pred, AggSpec, Partial, make_preds, make_aggs,   # POD shapes only
                       PRED_LT, PRED_LE, PRED_GT, PRED_GE, PRED_EQ, PRED_NE)

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB = os.path.join(_HERE, "liboracle.so")
if not os.path.exists(_LIB):
    raise ImportError("liboracle.so not built — run `make -C oracle`")

_lib = C.CDLL(_LIB)

_lib.oracle_open.restype = C.c_void_p
_lib.oracle_open.argtypes = [C.c_char_p]
_lib.oracle_close.argtypes = [C.c_void_p]
_lib.oracle_row_count.restype = C.c_uint64
_lib.oracle_row_count.argtypes = [C.c_void_p]
_lib.oracle_column_count.restype = C.c_uint32
_lib.oracle_column_count.argtypes = [C.c_void_p]
_lib.oracle_read_chunk.restype = C.c_int
_lib.oracle_read_chunk.argtypes = [C.c_void_p, C.c_uint32, C.c_uint32, C.c_uint32,
                                   C.c_void_p, C.c_void_p]
_lib.oracle_scan_agg_mt.restype = C.c_int
_lib.oracle_scan_agg_mt.argtypes = [C.c_void_p, C.POINTER(Pred), C.c_uint32,
                                    C.POINTER(AggSpec), C.c_uint32,
                                    C.POINTER(Partial), C.POINTER(C.c_int)]
_lib.oracle_scan_agg.restype = C.c_int
_lib.oracle_scan_agg.argtypes = [C.c_void_p, C.c_uint64, C.POINTER(Pred), C.c_uint32,
                                 C.POINTER(AggSpec), C.c_uint32,
                                 C.POINTER(C.c_uint32), C.c_uint32,
                                 C.POINTER(Partial), C.POINTER(C.c_uint32),
                                 C.POINTER(C.c_uint32), C.POINTER(C.c_int64)]


class OracleTable:
    def __init__(self, path):
        self._h = _lib.oracle_open(path.encode())
        if not self._h:
            raise RuntimeError(f"oracle_open({path}) failed")

    def close(self):
        if self._h:
            _lib.oracle_close(self._h)
            self._h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    @property
    def row_count(self):
        return _lib.oracle_row_count(self._h)

    @property
    def column_count(self):
        return _lib.oracle_column_count(self._h)

    def read_chunk(self, stripe, chunk, col, values_np, exists_np):
        rc = _lib.oracle_read_chunk(self._h, stripe, chunk, col,
                                    values_np.ctypes.data_as(C.c_void_p),
                                    exists_np.ctypes.data_as(C.c_void_p))
        if rc != 0:
            raise RuntimeError(f"oracle_read_chunk rc={rc}")

    def scan_agg(self, preds, aggs, group_cols=()):
        """Returns (partials, filtered) ungrouped, or
        ({(k0,k1): partials}, filtered) grouped."""
        parr = make_preds(preds)
        aarr = make_aggs(aggs)
        ngc = len(group_cols)
        gc = (C.c_uint32 * max(1, ngc))(*group_cols) if ngc else None
        cap = 64 if ngc else 1
        out = (Partial * (cap * len(aggs)))()
        keys = (C.c_uint32 * 64)()
        n_groups = C.c_uint32(0)
        filt = C.c_int64(0)
        rc = _lib.oracle_scan_agg(self._h, 0, parr, len(preds), aarr, len(aggs),
                                  gc, ngc, out, keys, C.byref(n_groups), C.byref(filt))
        if rc != 0:
            raise RuntimeError(f"oracle_scan_agg rc={rc}")
        if ngc:
            res = {}
            for g in range(n_groups.value):
                from citus_amd import decode_group_key
                key = decode_group_key(keys[g])
                res[key] = [out[g * len(aggs) + a] for a in range(len(aggs))]
            return res, filt.value
        return [out[a] for a in range(len(aggs))], filt.value

    def scan_agg_mt(self, preds, aggs):
        """all-core variant (OpenMP over chunk groups) -> (partials, cores)"""
        parr = make_preds(preds)
        aarr = make_aggs(aggs)
        out = (Partial * len(aggs))()
        cores = C.c_int(0)
        rc = _lib.oracle_scan_agg_mt(self._h, parr, len(preds), aarr, len(aggs),
                                     out, C.byref(cores))
        if rc != 0:
            raise RuntimeError(f"oracle_scan_agg_mt rc={rc}")
        return [out[a] for a in range(len(aggs))], cores.value
