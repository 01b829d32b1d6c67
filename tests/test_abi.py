"""
C-ABI boundary checks (include/cstripe.h): every declared symbol exported by
libcstripe.so; GPU entry points fail loudly without a device; combine
semantics (coord_combine_agg restatement) testable on CPU.
"""
import ctypes as C
import os
import re

import pytest

import citus_amd as ca

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_all_header_symbols_exported():
    hdr = open(os.path.join(REPO, "include", "cstripe.h")).read()
    # function declarations: name directly followed by '(' at top level
    names = re.findall(r"\b((?:cstripe|cagg|csbench)_\w+)\s*\(", hdr)
    lib = C.CDLL(os.path.join(REPO, "citus_amd", "libcstripe.so"))
    missing = [n for n in set(names) if not hasattr(lib, n)]
    # csbench_gen_lineitem is not in the header (bench tool), ignore inverse
    assert not missing, f"symbols declared but not exported: {missing}"
    assert hasattr(lib, "csbench_gen_lineitem")


def test_abi_version():
    assert C.CDLL(os.path.join(REPO, "citus_amd", "libcstripe.so")).cstripe_abi_version() == 2


def test_gpu_calls_fail_loudly_without_device(tmp_path):
    if ca.gpu_available():
        pytest.skip("GPU present")
    import numpy as np
    path = str(tmp_path / "t.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [np.arange(10, dtype=np.int64)])
    with ca.Reader(path) as r, r.scan(cols_mask=1) as s:
        with pytest.raises(ca.CStripeError, match="(?i)gpu|hip"):
            s.stage()
        with pytest.raises(ca.CStripeError, match="(?i)gpu|stage"):
            s.agg([(ca.AGG_COUNT_STAR, -1)])


def mkpart(**kw):
    p = ca.Partial()
    for k, v in kw.items():
        setattr(p, k, v)
    return p


def test_combine_strict_null_skip():
    """aggregate_utils.c:976-1000: strict combine skips NULL partials; the
    first non-NULL initializes."""
    aggs = [(ca.AGG_SUM_I64, 0)]
    out = ca.combine(aggs, [[mkpart(is_null=1)],
                            [mkpart(i128_lo=5, count=2)],
                            [mkpart(is_null=1)],
                            [mkpart(i128_lo=-3, i128_hi=-1, count=1)]])
    # {lo:-3, hi:-1} is two's-complement -3; 5 + (-3) = 2
    assert out[0].i128 == 2
    assert out[0].count == 3
    assert not out[0].is_null


def test_combine_all_null_sum_stays_null_count_coalesces():
    aggs = [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1)]
    out = ca.combine(aggs, [[mkpart(is_null=1), mkpart(is_null=1)],
                            [mkpart(is_null=1), mkpart(is_null=1)]])
    assert out[0].is_null                       # strict SUM stays NULL
    assert not out[1].is_null and out[1].count == 0   # COUNT -> COALESCE 0


def test_combine_minmax():
    aggs = [(ca.AGG_MIN_I64, 0), (ca.AGG_MAX_I64, 0)]
    parts = [[mkpart(i128_lo=-7, i128_hi=-1, count=1), mkpart(i128_lo=-7, i128_hi=-1, count=1)],
             [mkpart(i128_lo=100, count=1), mkpart(i128_lo=100, count=1)]]
    out = ca.combine(aggs, parts)
    assert out[0].i128 == -7
    assert out[1].i128 == 100


def test_combine_i128_carry():
    aggs = [(ca.AGG_SUM_I64, 0)]
    big = (1 << 63) - 1
    parts = [[mkpart(i128_lo=big, count=1)], [mkpart(i128_lo=big, count=1)],
             [mkpart(i128_lo=big, count=1)]]
    out = ca.combine(aggs, parts)
    assert out[0].i128 == 3 * big


def test_combine_f64_and_count_col():
    aggs = [(ca.AGG_SUM_F64, 0), (ca.AGG_COUNT_COL, 0)]
    parts = [[mkpart(f64=1.5, count=2), mkpart(count=2)],
             [mkpart(f64=-0.25, count=1), mkpart(count=1)]]
    out = ca.combine(aggs, parts)
    assert out[0].f64 == 1.25
    assert out[1].count == 3


def test_host_c_demo_compiles_and_links(tmp_path):
    """tools/q6_host.c — the plain-C consumer of the ABI — compiles with gcc
    (C, not C++) and fails loudly at stage without a GPU."""
    import subprocess
    exe = str(tmp_path / "q6_host")
    subprocess.check_call(
        ["gcc", "-O2", "-std=c11", os.path.join(REPO, "tools", "q6_host.c"),
         "-L" + os.path.join(REPO, "citus_amd"), "-lcstripe",
         "-Wl,-rpath," + os.path.join(REPO, "citus_amd"),
         "-I" + os.path.join(REPO, "include"), "-o", exe])
    r = subprocess.run([exe, os.path.join(REPO, "tests", "golden",
                                          "lineitem12k_lz4.cs")],
                       capture_output=True, text=True)
    assert "rows=12000" in r.stdout
    if not ca.gpu_available():
        assert r.returncode == 1 and "MI355X" in r.stderr
