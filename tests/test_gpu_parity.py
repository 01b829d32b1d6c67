"""
GPU parity tests (@pytest.mark.gpu — run on a real MI355X): the HIP path
(LZ4 decode kernel + fused filter/aggregate kernels), called through the
C ABI, against the CPU oracle on identical inputs.

Bar (BASELINE.json): bit-exact COUNT and integer/fixed-point SUM/MIN/MAX;
<= 1e-6 relative for float SUM (order of f64 additions differs).
"""
import os

import numpy as np
import pytest

import citus_amd as ca
import futil
import oracle
from test_format import read_all

from conftest import q6_preds

pytestmark = pytest.mark.gpu

RNG = np.random.default_rng(7)


@pytest.fixture(scope="module", autouse=True)
def require_gpu():
    assert ca.gpu_available(), "MI355X required for these tests"


def both(path, preds, aggs):
    with oracle.OracleTable(path) as t:
        op, ofilt = t.scan_agg(preds, aggs)
    with ca.Reader(path) as r, r.scan(cols_mask=ca.agg_cols_mask(aggs), preds=preds) as s:
        s.stage()
        gp = s.agg(aggs)
        gfilt = s.chunk_groups_filtered
    return op, ofilt, gp, gfilt


def assert_parity(op, gp, aggs):
    for i, a in enumerate(aggs):
        kind = a[0]
        assert op[i].is_null == gp[i].is_null, f"agg {i} null mismatch"
        assert op[i].count == gp[i].count, f"agg {i} count mismatch"
        if op[i].is_null:
            continue          # value fields are zeroed/undefined for NULL
        if kind == ca.AGG_SUM_F64 or kind in (ca.AGG_MIN_F64, ca.AGG_MAX_F64):
            if not op[i].is_null:
                if op[i].f64 == 0:
                    assert abs(gp[i].f64) < 1e-9
                else:
                    assert abs(gp[i].f64 - op[i].f64) <= 1e-6 * abs(op[i].f64)
        else:
            assert op[i].i128 == gp[i].i128, f"agg {i} i128 mismatch: {op[i].i128} vs {gp[i].i128}"


@pytest.mark.parametrize("variant", ["lz4", "none", "zstd"])
def test_q6_golden_fixture(golden_dir, expected, variant):
    path = os.path.join(golden_dir, f"lineitem12k_{variant}.cs")
    preds = q6_preds(ca, expected)
    aggs = [(ca.AGG_SUM_PROD_I64, 2, 3), (ca.AGG_COUNT_STAR, -1)]
    op, ofilt, gp, gfilt = both(path, preds, aggs)
    assert gp[0].i128 == expected["q6"]["revenue_scale4"]
    assert ofilt == gfilt
    assert_parity(op, gp, aggs)


@pytest.mark.parametrize("comp", [ca.COMP_NONE, ca.COMP_LZ4, ca.COMP_ZSTD,
                                  ca.COMP_PGLZ])
@pytest.mark.parametrize("n", [1, 999, 10000, 123457])
def test_agg_kinds_random(tmp_path, comp, n):
    a = RNG.integers(-10**6, 10**6, n).astype(np.int64)
    b = RNG.integers(0, 10**4, n).astype(np.int64)
    f = RNG.normal(size=n)
    path = str(tmp_path / f"r{comp}_{n}.cs")
    ca.write_table(path, [("a", ca.I64, 0), ("b", ca.I64, 0), ("f", ca.F64, 0)],
                   [a, b, f], compression=comp,
                   stripe_row_limit=50000, chunk_group_row_limit=5000)
    preds = [(0, ca.PRED_GT, -500000), (0, ca.PRED_LE, 700000)]
    aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_COUNT_COL, 1),
            (ca.AGG_SUM_I64, 0), (ca.AGG_SUM_I64, 1),
            (ca.AGG_MIN_I64, 0), (ca.AGG_MAX_I64, 0),
            (ca.AGG_SUM_F64, 2), (ca.AGG_MIN_F64, 2), (ca.AGG_MAX_F64, 2),
            (ca.AGG_SUM_PROD_I64, 0, 1)]
    op, ofilt, gp, gfilt = both(path, preds, aggs)
    assert ofilt == gfilt
    assert_parity(op, gp, aggs)


def test_nulls_parity(tmp_path):
    n = 34567
    a = RNG.integers(-1000, 1000, n).astype(np.int64)
    na = (RNG.random(n) < 0.35).astype(np.uint8)
    b = RNG.integers(0, 100, n).astype(np.int64)
    nb = (RNG.random(n) < 0.05).astype(np.uint8)
    path = str(tmp_path / "nulls.cs")
    ca.write_table(path, [("a", ca.I64, 0), ("b", ca.I64, 0)], [a, b],
                   nulls=[na, nb], compression=ca.COMP_LZ4,
                   chunk_group_row_limit=3000)
    preds = [(1, ca.PRED_GE, 10)]      # NULL b rows must fail the qual
    aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_COUNT_COL, 0),
            (ca.AGG_SUM_I64, 0), (ca.AGG_MIN_I64, 0), (ca.AGG_MAX_I64, 0),
            (ca.AGG_SUM_PROD_I64, 0, 1)]
    op, ofilt, gp, gfilt = both(path, preds, aggs)
    assert_parity(op, gp, aggs)


def test_all_null_column(tmp_path):
    n = 5000
    a = np.zeros(n, dtype=np.int64)
    na = np.ones(n, dtype=np.uint8)
    path = str(tmp_path / "allnull.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [a], nulls=[na],
                   compression=ca.COMP_LZ4)
    aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_COUNT_COL, 0), (ca.AGG_SUM_I64, 0)]
    op, _, gp, _ = both(path, [], aggs)
    assert gp[0].count == n
    assert gp[1].count == 0
    assert gp[2].is_null
    assert_parity(op, gp, aggs)


def test_pruning_parity(tmp_path):
    """simple_chunk_filtering vector on device: count 111111, 12 pruned."""
    n = 234568
    arr = np.arange(n, dtype=np.int64)
    path = str(tmp_path / "simple.cs")
    ca.write_table(path, [("i", ca.I64, 0)], [arr],
                   compression=ca.COMP_LZ4, chunk_group_row_limit=10000)
    with ca.Reader(path) as r, r.scan(preds=[(0, ca.PRED_GT, 123456)]) as s:
        s.stage()
        gp = s.agg([(ca.AGG_COUNT_STAR, -1)])
        assert gp[0].count == 111111
        assert s.chunk_groups_filtered == 12


def test_decode_parity_batches(tmp_path):
    """GPU LZ4 decode byte-parity: next_batch row-aligned output equals the
    oracle's DeserializeChunkData restatement for every chunk."""
    n = 47000
    a = RNG.integers(-2**40, 2**40, n).astype(np.int64)
    na = (RNG.random(n) < 0.2).astype(np.uint8)
    b = RNG.normal(size=n).astype(np.float32)
    path = str(tmp_path / "dec.cs")
    ca.write_table(path, [("a", ca.I64, 0), ("b", ca.F32, 0)], [a, b],
                   nulls=[na, None], compression=ca.COMP_LZ4,
                   stripe_row_limit=20000, chunk_group_row_limit=3000)

    with ca.Reader(path) as r, r.scan(cols_mask=0b11) as s:
        s.stage()
        va = np.zeros(3000, dtype=np.int64)
        vb = np.zeros(3000, dtype=np.float32)
        ea = np.zeros(3000, dtype=np.uint8)
        eb = np.zeros(3000, dtype=np.uint8)
        ova = np.zeros(3000, dtype=np.int64)
        ovb = np.zeros(3000, dtype=np.float32)
        oea = np.zeros(3000, dtype=np.uint8)
        oeb = np.zeros(3000, dtype=np.uint8)
        with oracle.OracleTable(path) as t:
            stripe = chunk = 0
            rows_left_in_stripe = min(20000, n)
            seen = 0
            while True:
                res = s.next_batch({0: va, 1: vb}, {0: ea, 1: eb})
                if res is None:
                    break
                nr, first = res
                t.read_chunk(stripe, chunk, 0, ova, oea)
                t.read_chunk(stripe, chunk, 1, ovb, oeb)
                np.testing.assert_array_equal(va[:nr], ova[:nr])
                np.testing.assert_array_equal(vb[:nr], ovb[:nr])
                np.testing.assert_array_equal(ea[:nr], 1 - oea[:nr])  # nulls vs exists
                seen += nr
                chunk += 1
                rows_left_in_stripe -= nr
                if rows_left_in_stripe == 0:
                    stripe += 1
                    chunk = 0
                    rows_left_in_stripe = min(20000, n - seen)
            assert seen == n


def test_repeated_agg_calls_stable(golden_dir, expected):
    """rescan semantics: re-invoking scan_agg re-runs decode+filter+agg and
    returns identical results."""
    path = os.path.join(golden_dir, "lineitem12k_lz4.cs")
    preds = q6_preds(ca, expected)
    aggs = [(ca.AGG_SUM_PROD_I64, 2, 3)]
    with ca.Reader(path) as r, r.scan(cols_mask=ca.agg_cols_mask(aggs), preds=preds) as s:
        s.stage()
        r1 = s.agg(aggs)
        r2 = s.agg(aggs)
        assert r1[0].i128 == r2[0].i128 == expected["q6"]["revenue_scale4"]
        assert s.last_kernel_ms > 0


def test_q6_large_synthetic(tmp_path):
    """config-1-shaped: 1M-row lineitem, count(*) WHERE l_quantity < 24 —
    plus full Q6 aggregate; GPU vs oracle bit-exact."""
    path = str(tmp_path / "li1m.cs")
    ca.gen_lineitem(path, 1_000_000)
    preds_c1 = [(1, ca.PRED_LT, 2400)]
    aggs_c1 = [(ca.AGG_COUNT_STAR, -1)]
    op, _, gp, _ = both(path, preds_c1, aggs_c1)
    assert_parity(op, gp, aggs_c1)

    preds_q6 = [(5, ca.PRED_GE, 8766), (5, ca.PRED_LT, 9131),
                (3, ca.PRED_GE, 5), (3, ca.PRED_LE, 7), (1, ca.PRED_LT, 2400)]
    aggs_q6 = [(ca.AGG_SUM_PROD_I64, 2, 3), (ca.AGG_COUNT_STAR, -1)]
    op, ofilt, gp, gfilt = both(path, preds_q6, aggs_q6)
    assert ofilt == gfilt
    assert_parity(op, gp, aggs_q6)


@pytest.mark.parametrize("seg", [("bytes", 0), ("kb", 8), ("kb", 1024), ("canon", 0)])
def test_decode_kernel_variants(tmp_path, seg):
    """All decode paths give identical, oracle-exact results: lane-parallel
    (greedy 256B micro-segments), wave-cooperative LDS (8KB segments), global
    fallback (single whole-chunk block), and the canonical closed-form path
    (default writer)."""
    kind, val = seg
    path = str(tmp_path / f"v{kind}{val}.cs")
    kw = {"seg_kb": val} if kind == "kb" else {}
    if kind != "canon":
        kw["canonical"] = 0          # force the greedy parse + decode kernels
    ca.gen_lineitem(path, 300_000, **kw)
    preds = [(5, ca.PRED_GE, 8766), (5, ca.PRED_LT, 9131),
             (3, ca.PRED_GE, 5), (3, ca.PRED_LE, 7), (1, ca.PRED_LT, 2400)]
    aggs = [(ca.AGG_SUM_PROD_I64, 2, 3), (ca.AGG_COUNT_STAR, -1),
            (ca.AGG_MIN_I64, 2), (ca.AGG_MAX_I64, 2)]
    op, ofilt, gp, gfilt = both(path, preds, aggs)
    assert ofilt == gfilt
    assert_parity(op, gp, aggs)


def q1_aggs():
    return [(ca.AGG_SUM_I64, 1), (ca.AGG_SUM_I64, 2),
            (ca.AGG_SUM_DISC_I64, 2, 3, -1, 100),
            (ca.AGG_SUM_DISC_TAX_I64, 2, 3, 4, 100),
            (ca.AGG_COUNT_STAR, -1)]


@pytest.mark.parametrize("variant", ["lz4", "none"])
def test_q1_grouped_golden(golden_dir, expected, variant):
    """TPC-H Q1 GROUP BY (returnflag, linestatus) on device vs the
    reference's expected result table (multi_tpch_query1.out)."""
    path = os.path.join(golden_dir, f"lineitem12k_{variant}.cs")
    q1 = expected["q1"]
    aggs = q1_aggs()
    preds = [(5, ca.PRED_LE, q1["shipdate_le"])]
    with ca.Reader(path) as r, \
         r.scan(cols_mask=ca.agg_cols_mask(aggs) | (1 << 6) | (1 << 7),
                preds=preds) as s:
        s.stage()
        res = s.agg_grouped(aggs, (6, 7))
    rf = expected["flag_codes"]["returnflag"]
    ls = expected["flag_codes"]["linestatus"]
    assert len(res) == len(q1["groups"])
    for key, vals in q1["groups"].items():
        a, b = key.split(",")
        got = res[(rf[a], ls[b])]
        assert [got[0].i128, got[1].i128, got[2].i128, got[3].i128,
                got[4].count] == vals
    # and vs oracle partial-for-partial
    with oracle.OracleTable(path) as t:
        ores, _ = t.scan_agg(preds, aggs, group_cols=(6, 7))
    for k, parts in ores.items():
        for i in range(len(aggs)):
            assert res[k][i].i128 == parts[i].i128
            assert res[k][i].count == parts[i].count


def test_grouped_random_many_groups(tmp_path):
    """grouped parity on random data with ~40 distinct composite keys,
    nullable measures, mixed agg kinds."""
    n = 123_456
    k0 = RNG.integers(0, 8, n).astype(np.int8)
    k1 = RNG.integers(0, 5, n).astype(np.int8)
    v = RNG.integers(-10**6, 10**6, n).astype(np.int64)
    nv = (RNG.random(n) < 0.2).astype(np.uint8)
    f = RNG.normal(size=n)
    path = str(tmp_path / "grp.cs")
    ca.write_table(path, [("k0", ca.I8, 0), ("k1", ca.I8, 0),
                          ("v", ca.I64, 0), ("f", ca.F64, 0)],
                   [k0, k1, v, f], nulls=[None, None, nv, None],
                   compression=ca.COMP_LZ4, chunk_group_row_limit=3000)
    aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_I64, 2),
            (ca.AGG_MIN_I64, 2), (ca.AGG_MAX_I64, 2), (ca.AGG_SUM_F64, 3)]
    preds = [(2, ca.PRED_GT, -900000)]
    with ca.Reader(path) as r, \
         r.scan(cols_mask=ca.agg_cols_mask(aggs) | 0b11, preds=preds) as s:
        s.stage()
        res = s.agg_grouped(aggs, (0, 1))
    with oracle.OracleTable(path) as t:
        ores, _ = t.scan_agg(preds, aggs, group_cols=(0, 1))
    assert set(res.keys()) == set(ores.keys())
    assert len(res) == 40
    for k in ores:
        for i, (kind, *_rest) in enumerate(aggs):
            assert res[k][i].count == ores[k][i].count, (k, i)
            assert res[k][i].is_null == ores[k][i].is_null
            if kind == ca.AGG_SUM_F64:
                if not ores[k][i].is_null and ores[k][i].f64 != 0:
                    assert abs(res[k][i].f64 - ores[k][i].f64) <= 1e-6 * abs(ores[k][i].f64)
            else:
                assert res[k][i].i128 == ores[k][i].i128, (k, i)


def test_grouped_null_keys(tmp_path):
    """NULL group keys form their own group, like the reference's
    HashAggregate (grouping treats NULLs as equal) — round-1 limit removed."""
    n = 12_345
    k = (np.arange(n) % 3).astype(np.int8)
    nk = (np.arange(n) % 7 == 0).astype(np.uint8)     # NULL key rows
    v = np.arange(n, dtype=np.int64)
    path = str(tmp_path / "nk.cs")
    ca.write_table(path, [("k", ca.I8, 0), ("v", ca.I64, 0)], [k, v],
                   nulls=[nk, None], compression=ca.COMP_LZ4,
                   chunk_group_row_limit=2000)
    aggs = [(ca.AGG_SUM_I64, 1), (ca.AGG_COUNT_STAR, -1)]
    with ca.Reader(path) as r, \
         r.scan(cols_mask=ca.agg_cols_mask(aggs) | 1) as s:
        s.stage()
        res = s.agg_grouped(aggs, (0,))
    with oracle.OracleTable(path) as t:
        ores, _ = t.scan_agg([], aggs, group_cols=(0,))
    assert set(res) == set(ores)
    assert any(key[0] is None for key in res)
    for key in ores:
        assert res[key][0].i128 == ores[key][0].i128
        assert res[key][1].count == ores[key][1].count
    # independent expectation for the NULL group
    nmask = nk == 1
    nkey = (None, 0)
    assert res[nkey][0].i128 == int(v[nmask].sum())
    assert res[nkey][1].count == int(nmask.sum())


def test_decode_torture_patterns(tmp_path):
    """Adversarial LZ4 structures through the decode kernels: long RLE runs
    (match len >> offset), incompressible random bytes (literal-heavy),
    alternating patterns, tiny repeating dictionaries."""
    n = 200_000
    cases = {
        "rle": np.zeros(n, dtype=np.int64),                       # one giant run
        "rand": RNG.integers(-2**62, 2**62, n).astype(np.int64),  # incompressible
        "alt": np.tile(np.array([7, -7], dtype=np.int64), n // 2),
        "dict": RNG.integers(0, 3, n).astype(np.int64),           # tiny dictionary
        "ramp": np.arange(n, dtype=np.int64) * 977,
    }
    for name, a in cases.items():
        path = str(tmp_path / f"t_{name}.cs")
        ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_LZ4)
        aggs = [(ca.AGG_SUM_I64, 0), (ca.AGG_MIN_I64, 0), (ca.AGG_MAX_I64, 0),
                (ca.AGG_COUNT_STAR, -1)]
        op, _, gp, _ = both(path, [], aggs)
        assert_parity(op, gp, aggs)
        # and byte-exact decode through next_batch for one chunk
        with ca.Reader(path) as r, r.scan(cols_mask=1) as s:
            s.stage()
            v = np.zeros(10000, dtype=np.int64)
            e = np.zeros(10000, dtype=np.uint8)
            res = s.next_batch({0: v}, {0: e})
            assert res is not None
            nr, _ = res
            np.testing.assert_array_equal(v[:nr], a[:nr], err_msg=name)


def test_mixed_width_types_gpu(tmp_path):
    """i8/i16/i32/f32 columns through decode + filter + aggregate."""
    n = 60_000
    c16 = RNG.integers(-30000, 30000, n).astype(np.int16)
    c32 = RNG.integers(-2**30, 2**30, n).astype(np.int32)
    f32 = RNG.normal(size=n).astype(np.float32)
    c8 = RNG.integers(-100, 100, n).astype(np.int8)
    path = str(tmp_path / "mix.cs")
    ca.write_table(path, [("a", ca.I16, 0), ("b", ca.I32, 0),
                          ("c", ca.F32, 0), ("d", ca.I8, 0)],
                   [c16, c32, f32, c8], compression=ca.COMP_LZ4,
                   chunk_group_row_limit=7000)
    preds = [(0, ca.PRED_GT, -20000), (2, ca.PRED_LT, 1.5)]
    aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_I64, 1),
            (ca.AGG_MIN_I64, 3), (ca.AGG_MAX_I64, 0),
            (ca.AGG_SUM_F64, 2), (ca.AGG_MIN_F64, 2)]
    op, ofilt, gp, gfilt = both(path, preds, aggs)
    assert ofilt == gfilt
    assert_parity(op, gp, aggs)


def test_shard_split_combine_property(tmp_path):
    """Size-independent property at bench scale: scanning two disjoint shard
    files and combining partials (the coordinator-merge contract) equals the
    bit-exact expected values computable from the generator — and COUNT from
    the two halves sums exactly to the total row count with no predicate."""
    n = 10_000_000
    a = str(tmp_path / "sh0.cs")
    b = str(tmp_path / "sh1.cs")
    ca.gen_lineitem(a, n, seed=42)
    ca.gen_lineitem(b, n, seed=777)
    preds = [(5, ca.PRED_GE, 8766), (5, ca.PRED_LT, 9131),
             (3, ca.PRED_GE, 5), (3, ca.PRED_LE, 7), (1, ca.PRED_LT, 2400)]
    aggs = [(ca.AGG_SUM_PROD_I64, 2, 3), (ca.AGG_COUNT_STAR, -1),
            (ca.AGG_SUM_I64, 2), (ca.AGG_MIN_I64, 2), (ca.AGG_MAX_I64, 2)]
    partials = []
    counts = []
    for path in (a, b):
        with ca.Reader(path) as r, \
             r.scan(cols_mask=ca.agg_cols_mask(aggs), preds=preds) as s:
            s.stage()
            parts = s.agg(aggs)
            partials.append(parts)
        with ca.Reader(path) as r, r.scan(cols_mask=1 << 2) as s:
            s.stage()
            counts.append(s.agg([(ca.AGG_COUNT_STAR, -1)])[0].count)
    assert counts[0] == counts[1] == n          # unfiltered count == rows
    combined = ca.combine(aggs, partials)
    # oracle cross-check of the combined result (per-shard partial + combine
    # on CPU — the worker/coordinator split applied identically)
    oparts = []
    for path in (a, b):
        with oracle.OracleTable(path) as t:
            p, _ = t.scan_agg(preds, aggs)
            oparts.append(p)
    ocombined = ca.combine(aggs, oparts)
    for i in range(len(aggs)):
        assert combined[i].i128 == ocombined[i].i128
        assert combined[i].count == ocombined[i].count


def test_fuzz_differential():
    """Randomized differential fuzz: random schemas / sizes / null patterns /
    codecs / predicates / aggregates — GPU vs oracle, 24 cases."""
    TYPES = [(ca.I8, np.int8), (ca.I16, np.int16), (ca.I32, np.int32),
             (ca.I64, np.int64), (ca.F32, np.float32), (ca.F64, np.float64)]
    INT_AGGS = [ca.AGG_SUM_I64, ca.AGG_MIN_I64, ca.AGG_MAX_I64, ca.AGG_COUNT_COL]
    FLT_AGGS = [ca.AGG_SUM_F64, ca.AGG_MIN_F64, ca.AGG_MAX_F64, ca.AGG_COUNT_COL]
    import tempfile
    for seed in range(24):
        rng = np.random.default_rng(1000 + seed)
        ncols = int(rng.integers(1, 6))
        kinds = [TYPES[int(rng.integers(0, 6))] for _ in range(ncols)]
        n = int(rng.integers(1, 60001))
        chunk = int(rng.integers(1, 11)) * 1000
        stripe = [10000, 50000, 150000][int(rng.integers(0, 3))]
        comp = [ca.COMP_NONE, ca.COMP_LZ4, ca.COMP_LZ4, ca.COMP_ZSTD][int(rng.integers(0, 4))]
        canon = int(rng.integers(0, 2))   # both writer parses fuzzed
        cols, nulls, defs = [], [], []
        for i, (t, dt) in enumerate(kinds):
            if dt in (np.float32, np.float64):
                a = rng.normal(size=n).astype(dt)
            else:
                info = np.iinfo(dt)
                a = rng.integers(info.min // 2, info.max // 2, n).astype(dt)
            cols.append(a)
            p = float(rng.random() * 0.5) if rng.random() < 0.4 else 0.0
            nulls.append((rng.random(n) < p).astype(np.uint8) if p else None)
            defs.append((f"c{i}", t, 0))
        with tempfile.TemporaryDirectory() as td:
            path = os.path.join(td, "f.cs")
            ca.write_table(path, defs, cols, nulls=nulls, compression=comp,
                           stripe_row_limit=stripe, chunk_group_row_limit=chunk,
                           canonical=canon)
            preds = []
            for _ in range(int(rng.integers(0, 4))):
                ci = int(rng.integers(0, ncols))
                op = int(rng.integers(0, 6))
                t, dt = kinds[ci]
                if dt in (np.float32, np.float64):
                    preds.append((ci, op, float(rng.normal())))
                else:
                    preds.append((ci, op, int(rng.integers(-100, 100))))
            aggs = [(ca.AGG_COUNT_STAR, -1)]
            for ci, (t, dt) in enumerate(kinds):
                pool = FLT_AGGS if dt in (np.float32, np.float64) else INT_AGGS
                aggs.append((pool[int(rng.integers(0, len(pool)))], ci))
            op_, ofilt, gp_, gfilt = both(path, preds, aggs)
            assert ofilt == gfilt, f"seed {seed}"
            try:
                assert_parity(op_, gp_, aggs)
            except AssertionError as e:
                raise AssertionError(f"fuzz seed {seed}: {e}")


def test_shard_directory_scan(tmp_path):
    """config-3 shape: a directory of shard files scanned as ONE table (the
    static shard-group-per-GPU mapping) equals the combined per-shard oracle
    partials."""
    d = tmp_path / "shards"
    d.mkdir()
    for i in range(4):
        ca.gen_lineitem(str(d / f"shard{i:02d}.cs"), 500_000, seed=42 + i)
    preds = [(5, ca.PRED_GE, 8766), (5, ca.PRED_LT, 9131),
             (3, ca.PRED_GE, 5), (3, ca.PRED_LE, 7), (1, ca.PRED_LT, 2400)]
    aggs = [(ca.AGG_SUM_PROD_I64, 2, 3), (ca.AGG_COUNT_STAR, -1),
            (ca.AGG_MIN_I64, 2), (ca.AGG_MAX_I64, 2)]
    with ca.Reader(str(d)) as r:
        assert r.row_count == 2_000_000
        with r.scan(cols_mask=ca.agg_cols_mask(aggs), preds=preds) as s:
            s.stage()
            gp = s.agg(aggs)
    oparts = []
    for i in range(4):
        with oracle.OracleTable(str(d / f"shard{i:02d}.cs")) as t:
            p, _ = t.scan_agg(preds, aggs)
            oparts.append(p)
    oc = ca.combine(aggs, oparts)
    for i in range(len(aggs)):
        assert gp[i].i128 == oc[i].i128
        assert gp[i].count == oc[i].count


def test_fuzz_grouped_differential():
    """Randomized grouped differential fuzz: random i8 key columns, group
    counts spanning the fused 16-slot cap (exercises fused-grouped AND the
    >16-group fallback), nullable measures."""
    import tempfile
    for seed in range(12):
        rng = np.random.default_rng(5000 + seed)
        n = int(rng.integers(1000, 80001))
        nk = int(rng.integers(1, 3))
        kr = [int(rng.integers(2, 9)) for _ in range(nk)]   # up to 64 combos
        keys = [rng.integers(0, kr[i], n).astype(np.int8) for i in range(nk)]
        v = rng.integers(-10**6, 10**6, n).astype(np.int64)
        nv = (rng.random(n) < float(rng.random() * 0.4)).astype(np.uint8)
        comp = [ca.COMP_NONE, ca.COMP_LZ4, ca.COMP_LZ4][int(rng.integers(0, 3))]
        defs = [(f"k{i}", ca.I8, 0) for i in range(nk)] + [("v", ca.I64, 0)]
        cols = keys + [v]
        nulls = [None] * nk + [nv]
        with tempfile.TemporaryDirectory() as td:
            path = os.path.join(td, "g.cs")
            ca.write_table(path, defs, cols, nulls=nulls, compression=comp,
                           chunk_group_row_limit=int(rng.integers(1, 11)) * 1000,
                           canonical=int(rng.integers(0, 2)))
            preds = []
            if rng.random() < 0.6:
                preds.append((nk, int(rng.integers(0, 4)), int(rng.integers(-10**5, 10**5))))
            if rng.random() < 0.5:        # OR group over the measure column
                preds.append((nk, ca.PRED_LT, int(rng.integers(-10**5, 0)), 7))
                preds.append((nk, ca.PRED_GT, int(rng.integers(0, 10**5)), 7))
            aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_I64, nk),
                    (ca.AGG_MIN_I64, nk), (ca.AGG_MAX_I64, nk)]
            gcols = tuple(range(nk))
            mask = ca.agg_cols_mask(aggs)
            for c in gcols:
                mask |= 1 << c
            with ca.Reader(path) as r, r.scan(cols_mask=mask, preds=preds) as s:
                s.stage()
                res = s.agg_grouped(aggs, gcols)
            with oracle.OracleTable(path) as t:
                ores, _ = t.scan_agg(preds, aggs, group_cols=gcols)
            assert set(res.keys()) == set(ores.keys()), f"seed {seed}"
            for k in ores:
                for i in range(len(aggs)):
                    assert res[k][i].count == ores[k][i].count, (seed, k, i)
                    assert res[k][i].is_null == ores[k][i].is_null, (seed, k, i)
                    assert res[k][i].i128 == ores[k][i].i128, (seed, k, i)


def test_corrupted_stream_flags_error(tmp_path):
    """Flip bytes inside a compressed value stream: the decode kernel must
    flag the malformed block (or, at worst, decode to the declared size) —
    never crash or hang."""
    path = str(tmp_path / "c.cs")
    n = 50_000
    ca.write_table(path, [("a", ca.I64, 0)],
                   [RNG.integers(0, 1000, n).astype(np.int64)],
                   compression=ca.COMP_LZ4)
    raw = bytearray(open(path, "rb").read())
    # smash a 64-byte span in the middle of the data region
    mid = len(raw) // 3
    for i in range(64):
        raw[mid + i] ^= 0xA5
    bad = str(tmp_path / "bad.cs")
    open(bad, "wb").write(bytes(raw))
    try:
        with ca.Reader(bad) as r, r.scan(cols_mask=1) as s:
            s.stage()
            try:
                s.agg([(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1)])
                # rare: the flips may land outside selected streams or still
                # form a valid block — surviving without a crash is the bar
            except ca.CStripeError as e:
                assert "decode" in str(e).lower() or "format" in str(e).lower()
    except ca.CStripeError:
        pass   # footer-region corruption: clean open/stage failure also fine


def test_stage_cycle_no_leak(tmp_path):
    """Repeated stage/end cycles: device allocations are released (a leak
    would OOM long before 120 iterations of ~60 MB)."""
    path = str(tmp_path / "l.cs")
    ca.gen_lineitem(path, 2_000_000)
    aggs = [(ca.AGG_COUNT_STAR, -1)]
    with ca.Reader(path) as r:
        for i in range(120):
            with r.scan(cols_mask=1 << 1, preds=[(1, ca.PRED_LT, 2400)]) as s:
                s.stage()
                parts = s.agg(aggs)
                assert parts[0].count > 0


def test_tail_segment_fused_parity(tmp_path):
    """Advisor r1 (high): rows % 32 == 1 dense i64 chunks produce an 8 B final
    LZ4 segment; the fused kernel must aggregate them exactly (the old writer
    absorbed the tail into a 264 B segment the kernel misread)."""
    for n in (10001, 9985, 150001):
        a = ((np.arange(n, dtype=np.int64) * 2654435761) % 5000).astype(np.int64)
        b = ((np.arange(n, dtype=np.int64) * 40503) % 9000).astype(np.int64)
        path = str(tmp_path / f"tail{n}.cs")
        ca.write_table(path, [("a", ca.I64, 0), ("b", ca.I64, 0)], [a, b],
                       compression=ca.COMP_LZ4)
        preds = [(0, ca.PRED_LT, 2400)]
        aggs = [(ca.AGG_SUM_PROD_I64, 0, 1), (ca.AGG_COUNT_STAR, -1),
                (ca.AGG_SUM_I64, 1)]
        op, ofilt, gp, gfilt = both(path, preds, aggs)
        assert ofilt == gfilt
        assert_parity(op, gp, aggs)
        mask = a < 2400
        assert gp[0].i128 == int((a[mask].astype(object) * b[mask].astype(object)).sum())


def test_nan_semantics_parity(tmp_path):
    """PG float ordering: NaN sorts above every value and equals itself —
    predicate eval, chunk pruning and MIN/MAX must all agree with the oracle
    (which is pinned to float8_cmp_internal semantics)."""
    n = 4000
    a = RNG.normal(size=n) * 100
    a[::7] = np.nan
    path = str(tmp_path / "nan.cs")
    ca.write_table(path, [("a", ca.F64, 0)], [a], compression=ca.COMP_LZ4,
                   chunk_group_row_limit=500)
    for preds in ([(0, ca.PRED_GT, 50.0)], [(0, ca.PRED_LT, 50.0)],
                  [(0, ca.PRED_GE, 1e12)], []):
        aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_MIN_F64, 0), (ca.AGG_MAX_F64, 0),
                (ca.AGG_SUM_F64, 0)]
        op, ofilt, gp, gfilt = both(path, preds, aggs)
        assert ofilt == gfilt
        assert op[0].count == gp[0].count
        for i in (1, 2):
            if np.isnan(op[i].f64):
                assert np.isnan(gp[i].f64)
            else:
                assert op[i].f64 == gp[i].f64
    # col > huge: only NaN rows pass; count must be exactly the NaN count
    op, _, gp, _ = both(path, [(0, ca.PRED_GT, 1e12)], [(ca.AGG_COUNT_STAR, -1)])
    assert gp[0].count == op[0].count == int(np.isnan(a).sum())


def test_or_pushdown_golden_gpu(tmp_path):
    """OR pushdown on device vs the reference's expected chunk-filter vectors
    (expected/columnar_chunk_filtering.out:837-930) and the oracle."""
    a = np.arange(1, 200001, dtype=np.int64)
    path = str(tmp_path / "pushdown_test.cs")
    ca.write_table(path, [("a", ca.I64, 0)], [a], compression=ca.COMP_LZ4,
                   stripe_row_limit=2000, chunk_group_row_limit=1000)
    cases = [
        ([(0, ca.PRED_EQ, 204356, 1), (0, ca.PRED_EQ, 104356, 1),
          (0, ca.PRED_EQ, 76556, 1)], 2, 198, 180912),
        ([(0, ca.PRED_EQ, 194356, 1), (0, ca.PRED_EQ, 104356, 1),
          (0, ca.PRED_EQ, 76556, 1)], 3, 197, 375268),
        ([(0, ca.PRED_GT, 1000, 1), (0, ca.PRED_GT, 20000, 1),
          (0, ca.PRED_GT, 1000, 2), (0, ca.PRED_LT, 50000, 2),
          (0, ca.PRED_LT, 10000, 3), (0, ca.PRED_GT, 20000, 3),
          (0, ca.PRED_LT, 10000, 4), (0, ca.PRED_LT, 50000, 4)],
         38998, 161, 1099459500),
    ]
    aggs = [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1)]
    for preds, exp_rows, exp_filtered, exp_sum in cases:
        op, ofilt, gp, gfilt = both(path, preds, aggs)
        assert gfilt == ofilt == exp_filtered
        assert gp[1].count == exp_rows
        assert gp[0].i128 == exp_sum
        assert_parity(op, gp, aggs)


def test_or_pushdown_fuzz_gpu(tmp_path):
    """randomized CNF filters (mixed OR groups + standalone conjuncts,
    canonical and greedy writers) — GPU vs oracle."""
    rng = np.random.default_rng(99)
    for trial in range(8):
        n = int(rng.integers(5000, 60001))
        a = rng.integers(0, 5000, n).astype(np.int64)
        b = rng.integers(-1000, 1000, n).astype(np.int64)
        path = str(tmp_path / f"orf{trial}.cs")
        ca.write_table(path, [("a", ca.I64, 0), ("b", ca.I64, 0)], [a, b],
                       compression=ca.COMP_LZ4,
                       chunk_group_row_limit=int(rng.integers(1, 8)) * 1000,
                       canonical=int(rng.integers(0, 2)))
        preds = []
        for _ in range(int(rng.integers(1, 7))):
            preds.append((int(rng.integers(0, 2)), int(rng.integers(0, 6)),
                          int(rng.integers(-1200, 5200)),
                          int(rng.integers(0, 4))))
        aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_I64, 0), (ca.AGG_MIN_I64, 1)]
        op, ofilt, gp, gfilt = both(path, preds, aggs)
        assert ofilt == gfilt, (trial, preds)
        assert_parity(op, gp, aggs)


def test_text_flags_grouped_gpu(tmp_path):
    """Real char(1) varlena flag columns (round-1 VERDICT #6): generator
    lineitem now stores l_returnflag/l_linestatus as reference-layout
    short-varlena slots; GROUP BY them on device vs oracle and vs the
    generator-exact expected values."""
    n = 400_000
    path = str(tmp_path / "li.cs")
    ca.gen_lineitem(path, n, seed=42)
    aggs = [(ca.AGG_SUM_I64, 1), (ca.AGG_SUM_I64, 2),
            (ca.AGG_SUM_DISC_I64, 2, 3, -1, 100),
            (ca.AGG_SUM_DISC_TAX_I64, 2, 3, 4, 100),
            (ca.AGG_COUNT_STAR, -1)]
    preds = [(5, ca.PRED_LE, 10471)]
    mask = ca.agg_cols_mask(aggs) | (1 << 6) | (1 << 7)
    with ca.Reader(path) as r, r.scan(cols_mask=mask, preds=preds) as s:
        assert r.column_def(6)[1] == ca.TEXT
        s.stage()
        res = s.agg_grouped(aggs, (6, 7))
    with oracle.OracleTable(path) as t:
        ores, _ = t.scan_agg(preds, aggs, group_cols=(6, 7))
    assert set(res) == set(ores)
    for k in ores:
        for i in range(len(aggs)):
            assert res[k][i].i128 == ores[k][i].i128, (k, i)
            assert res[k][i].count == ores[k][i].count
    RF = {0: ord("A"), 1: ord("N"), 2: ord("R")}
    LS = {0: ord("O"), 1: ord("F")}
    exp = {(RF[a], LS[b]): v for (a, b), v in ca.expected_q1(n, 42).items()}
    assert set(res) == set(exp)
    for k, v in exp.items():
        got = res[k]
        assert [got[0].i128, got[1].i128, got[2].i128, got[3].i128,
                got[4].count] == v


def test_text_pred_gpu(tmp_path):
    n = 50_000
    flags = np.array([["A", "N", "R"][i % 3] for i in range(n)])
    slots = ca.text_slots(flags)
    v = np.arange(n, dtype=np.int64)
    path = str(tmp_path / "tp.cs")
    ca.write_table(path, [("f", ca.TEXT, 0), ("v", ca.I64, 0)], [slots, v],
                   compression=ca.COMP_LZ4)
    preds = [(0, ca.PRED_EQ, ca.text_slot("N"))]
    aggs = [(ca.AGG_SUM_I64, 1), (ca.AGG_COUNT_STAR, -1)]
    op, ofilt, gp, gfilt = both(path, preds, aggs)
    assert ofilt == gfilt
    assert_parity(op, gp, aggs)
    m = flags == "N"
    assert gp[0].i128 == int(v[m].sum())


def test_read_row_random_access(tmp_path):
    """cstripe_read_row (ColumnarReadRowByRowNumber, columnar_reader.c:
    386-441): point lookups across chunk/stripe boundaries, nulls, TEXT
    slots, and the no-such-row / predicate-rejection contracts."""
    n = 25_000
    a = (np.arange(n, dtype=np.int64) * 13) % 100_000
    f = ca.text_slots([["A", "N", "R"][i % 3] for i in range(n)])
    na = (np.arange(n) % 9 == 0).astype(np.uint8)
    path = str(tmp_path / "rr.cs")
    ca.write_table(path, [("a", ca.I64, 0), ("f", ca.TEXT, 0)], [a, f],
                   nulls=[na, None], compression=ca.COMP_LZ4,
                   stripe_row_limit=8000, chunk_group_row_limit=1000)
    with ca.Reader(path) as r, r.scan(cols_mask=0b11) as s:
        s.stage()
        for row in (0, 1, 999, 1000, 7999, 8000, 13333, n - 1):
            got = s.read_row(row)
            if na[row]:
                assert got[0] is None
            else:
                assert got[0] == int(a[row]), row
            assert got[1] == int(f[row]), row
        assert s.read_row(n) is None            # no such row -> END/None
        assert s.read_row(2) is not None        # cache revisit
    with ca.Reader(path) as r, r.scan(cols_mask=0b11,
                                      preds=[(0, ca.PRED_GT, 10)]) as s:
        s.stage()
        with pytest.raises(ca.CStripeError, match="predicate-free"):
            s.read_row(5)


def test_device_write_path(tmp_path):
    """cstripe_write_rows_device (SURVEY §8f4 write side): HBM-resident
    columns compressed ON the GPU into canonical streams; the file must be
    byte-compatible with the host writer's output semantics — the oracle
    (system liblz4) reads it back exactly, skip-node min/max match, and a
    GPU scan over it is parity-green."""
    import torch
    n = 123_456
    cpu_a = ((torch.arange(n, dtype=torch.int64) * 7919) % 5000)
    cpu_b = 1000 + ((torch.arange(n, dtype=torch.int64) * 104729) % 90000)
    cpu_f = torch.rand(n, dtype=torch.float64) * 100
    a, b, f = cpu_a.cuda(), cpu_b.cuda(), cpu_f.cuda()
    path = str(tmp_path / "dev.cs")
    defs = [("a", ca.I64, 0), ("b", ca.I64, 0), ("f", ca.F64, 0)]
    ca.write_table_device(path, defs,
                          [a.data_ptr(), b.data_ptr(), f.data_ptr()], n)
    torch.cuda.synchronize()

    foot = futil.read_footer(path)
    node_a = foot["stripes"][0]["nodes"][0][0]
    assert node_a["comp_type"] == ca.COMP_LZ4
    assert futil.SEGMODE_P_BASE < node_a["segs"][0]["mode"] <= futil.SEGMODE_P_BASE + 4
    an, bn, fn = cpu_a.numpy(), cpu_b.numpy(), cpu_f.numpy()
    assert node_a["min_i"] == int(an[:10000].min())
    assert node_a["max_i"] == int(an[:10000].max())

    with oracle.OracleTable(path) as t:
        assert t.row_count == n
        v, e = read_all(t, 0, n, np.int64, 10000, stripe_rows=150000)
        np.testing.assert_array_equal(v, an)
        v, _ = read_all(t, 2, n, np.float64, 10000, stripe_rows=150000)
        np.testing.assert_array_equal(v, fn)   # raw copy-back column

    preds = [(0, ca.PRED_LT, 2400)]
    aggs = [(ca.AGG_SUM_I64, 1), (ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_F64, 2)]
    op, ofilt, gp, gfilt = both(path, preds, aggs)
    assert ofilt == gfilt
    assert_parity(op, gp, aggs)


def test_device_write_matches_host_writer(tmp_path):
    """device and host writers produce files with identical decoded content
    and skip nodes for the same input (compressed-byte parity is unpinned,
    as with the reference's codecs)."""
    import torch
    n = 40_000
    vals = ((torch.arange(n, dtype=torch.int64) * 31) % 700)
    const = torch.full((n,), 42, dtype=torch.int64)
    pd = str(tmp_path / "d.cs")
    ph = str(tmp_path / "h.cs")
    dvals, dconst = vals.cuda(), const.cuda()   # keep refs: data_ptr lifetime
    ca.write_table_device(pd, [("v", ca.I64, 0), ("c", ca.I64, 0)],
                          [dvals.data_ptr(), dconst.data_ptr()], n)
    torch.cuda.synchronize()
    ca.write_table(ph, [("v", ca.I64, 0), ("c", ca.I64, 0)],
                   [vals.numpy(), const.numpy()])
    fd, fh = futil.read_footer(pd), futil.read_footer(ph)
    assert fd["stripes"][0]["nodes"][1][0]["segs"][0]["mode"] == futil.SEGMODE_CONST
    for t in (fd, fh):
        assert t["total_rows"] == n
    for path in (pd, ph):
        with oracle.OracleTable(path) as t:
            parts, _ = t.scan_agg([], [(ca.AGG_SUM_I64, 0), (ca.AGG_MIN_I64, 0),
                                       (ca.AGG_MAX_I64, 1)])
            assert parts[0].i128 == int(vals.sum())
            assert parts[1].i128 == int(vals.min())
            assert parts[2].i128 == 42


def test_device_write_edge_shapes(tmp_path):
    """device write: spans that do not align to chunks (host round trip),
    tiny tables, chunk/stripe boundary crossings, canonical=0."""
    import torch
    for n in (1, 2, 7, 9_999, 10_001, 150_001):
        t = ((torch.arange(n, dtype=torch.int64) * 13) % 300).cuda()
        path = str(tmp_path / f"e{n}.cs")
        ca.write_table_device(path, [("v", ca.I64, 0)], [t.data_ptr()], n)
        torch.cuda.synchronize()
        with oracle.OracleTable(path) as ot:
            assert ot.row_count == n
            parts, _ = ot.scan_agg([], [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1)])
            assert parts[0].i128 == int(t.sum().item())
            assert parts[1].count == n
        del t
    # canonical off: everything goes through the host greedy path
    n = 25_000
    t = ((torch.arange(n, dtype=torch.int64) * 7) % 50).cuda()
    path = str(tmp_path / "nc.cs")
    ca.write_table_device(path, [("v", ca.I64, 0)], [t.data_ptr()], n,
                          canonical=0)
    torch.cuda.synchronize()
    foot = futil.read_footer(path)
    assert all(sg["mode"] == futil.SEGMODE_GENERIC
               for sg in foot["stripes"][0]["nodes"][0][0]["segs"])
    with oracle.OracleTable(path) as ot:
        parts, _ = ot.scan_agg([], [(ca.AGG_SUM_I64, 0)])
        assert parts[0].i128 == int(t.sum().item())


def test_zstd_canonical_vs_generic_parity(tmp_path):
    """Canonical restricted-zstd frames (closed-form GPU reads, no FSE walk)
    must aggregate identically to the same data written with canonical=0
    (generic ZR frames, zr_decode kernel) and to the oracle."""
    n = 60000
    qty = ((RNG.integers(1, 51, n)) * 100).astype(np.int64)        # P(2)
    price = RNG.integers(90000, 200000, n).astype(np.int64)        # P(3)
    disc = RNG.integers(0, 11, n).astype(np.int64)                 # P(1)
    const = np.full(n, 7, dtype=np.int64)                          # CONST
    defs = [("q", ca.I64, 0), ("p", ca.I64, 0), ("d", ca.I64, 0),
            ("c", ca.I64, 0)]
    cols = [qty, price, disc, const]
    preds = [(0, ca.PRED_LT, 2400), (2, ca.PRED_GE, 1), (2, ca.PRED_LE, 6)]
    aggs = [(ca.AGG_SUM_PROD_I64, 1, 2), (ca.AGG_COUNT_STAR, -1),
            (ca.AGG_MIN_I64, 1), (ca.AGG_MAX_I64, 3)]
    results = []
    for canon in (1, 0):
        path = str(tmp_path / f"zc{canon}.cs")
        ca.write_table(path, defs, cols, compression=ca.COMP_ZSTD,
                       canonical=canon)
        foot = futil.read_footer(path)
        modes = {s["mode"] for nd_col in foot["stripes"][0]["nodes"]
                 for s in nd_col[0]["segs"]}
        if canon:
            assert any(m >= futil.SEGMODE_ZRP_BASE for m in modes), modes
        else:
            assert all(m == futil.SEGMODE_ZR for m in modes), modes
        op, ofilt, gp, gfilt = both(path, preds, aggs)
        assert ofilt == gfilt
        assert_parity(op, gp, aggs)
        results.append([(p.i128, p.count, p.is_null) for p in gp])
    assert results[0] == results[1]


def test_grouped_zstd_width4_canonical_flags(tmp_path):
    """Q1-shaped GROUP BY over char(1) short-varlena flag columns stored as
    width-4 canonical zstd frames (zr_canon_b4: one closed-form byte per
    row — no decode kernel): grouped partials must match the oracle and the
    same data written as lz4 (lane-decoded) bit-for-bit."""
    n = 50000
    rf = ca.text_slots(["A", "N", "R"][i % 3] for i in range(n))
    lst = ca.text_slots(["O", "F"][i % 2] for i in range(n))
    qty = ((RNG.integers(1, 51, n)) * 100).astype(np.int64)
    defs = [("rf", ca.TEXT, 0), ("ls", ca.TEXT, 0), ("q", ca.I64, 0)]
    cols = [rf.view(np.int32), lst.view(np.int32), qty]
    aggs = [(ca.AGG_SUM_I64, 2), (ca.AGG_COUNT_STAR, -1),
            (ca.AGG_MIN_I64, 2), (ca.AGG_MAX_I64, 2)]
    outs = {}
    for comp, name in [(ca.COMP_ZSTD, "zstd"), (ca.COMP_LZ4, "lz4")]:
        path = str(tmp_path / f"g4_{name}.cs")
        ca.write_table(path, defs, cols, compression=comp)
        if comp == ca.COMP_ZSTD:
            foot = futil.read_footer(path)
            m0 = foot["stripes"][0]["nodes"][0][0]["segs"][0]["mode"]
            assert m0 == futil.SEGMODE_ZR4B_BASE | 1, hex(m0)
        with oracle.OracleTable(path) as t:
            og, _ = t.scan_agg([], aggs, group_cols=(0, 1))
        with ca.Reader(path) as r, \
             r.scan(cols_mask=0b111, preds=[]) as s:
            s.stage()
            gg = s.agg_grouped(aggs, (0, 1))
        okeys = sorted(og.keys())
        gkeys = sorted(gg.keys())
        assert okeys == gkeys and len(okeys) == 6, name
        for k in okeys:
            for i in range(len(aggs)):
                assert og[k][i].i128 == gg[k][i].i128, (name, k, i)
                assert og[k][i].count == gg[k][i].count, (name, k, i)
        outs[name] = {k: [(p.i128, p.count) for p in v] for k, v in gg.items()}
    assert outs["zstd"] == outs["lz4"]


def test_batch_and_read_row_zstd_canonical(tmp_path):
    """canon_decode_kernel over CANONICAL zstd chunks (width-8 P and the
    width-4 single-varying-byte family): next_batch row-aligned output and
    random-access read_row must equal the oracle's view."""
    n = 26000
    q = ((RNG.integers(1, 51, n)) * 100).astype(np.int64)          # zstd P(2)
    fl = ca.text_slots(["A", "N", "R"][i % 3] for i in range(n))   # zr4b k=1
    path = str(tmp_path / "zb.cs")
    ca.write_table(path, [("q", ca.I64, 0), ("f", ca.TEXT, 0)],
                   [q, fl.view(np.int32)], compression=ca.COMP_ZSTD)
    foot = futil.read_footer(path)
    assert foot["stripes"][0]["nodes"][0][0]["segs"][0]["mode"] == 0x62
    assert foot["stripes"][0]["nodes"][1][0]["segs"][0]["mode"] == 0x81
    with ca.Reader(path) as r, r.scan(cols_mask=0b11) as s:
        s.stage()
        vq = np.zeros(10000, dtype=np.int64)
        vf = np.zeros(10000, dtype=np.int32)
        eq_ = np.zeros(10000, dtype=np.uint8)
        ef = np.zeros(10000, dtype=np.uint8)
        seen = 0
        while True:
            res = s.next_batch({0: vq, 1: vf}, {0: eq_, 1: ef})
            if res is None:
                break
            nr, first = res
            np.testing.assert_array_equal(vq[:nr], q[seen:seen + nr])
            np.testing.assert_array_equal(vf[:nr].view(np.uint32),
                                          fl[seen:seen + nr])
            assert not eq_[:nr].any() and not ef[:nr].any()   # 1 = NULL
            seen += nr
        assert seen == n
    with ca.Reader(path) as r, r.scan(cols_mask=0b11) as s:
        s.stage()
        for rn in (0, 1, 9999, 10000, 10001, 12345, n - 1):
            vals = s.read_row(rn)
            assert vals[0] == q[rn], rn
            assert (vals[1] & 0xFFFFFFFF) == int(fl[rn]), rn


def test_device_prune_matches_host(tmp_path):
    """chunk_prune_kernel (SelectedChunkMask on device) must remove EXACTLY
    the chunks the host CNF loop removes — OR groups, PG-NaN float ordering
    and all-NULL chunks included — and yield identical aggregates."""
    import os as _os
    n = 120_000
    a = RNG.integers(0, 10_000, n).astype(np.int64)
    f = RNG.normal(size=n)
    f[: n // 6] = np.nan                        # NaN-heavy early chunks
    b = RNG.integers(-500, 500, n).astype(np.int64)
    nb = np.zeros(n, dtype=np.uint8)
    nb[n // 3: n // 2] = 1                      # a run of all-NULL chunks
    path = str(tmp_path / "dp.cs")
    ca.write_table(path, [("a", ca.I64, 0), ("f", ca.F64, 0), ("b", ca.I64, 0)],
                   [np.sort(a), f, b], nulls=[None, None, nb],
                   compression=ca.COMP_LZ4, chunk_group_row_limit=1000,
                   stripe_row_limit=10000)
    aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_I64, 0), (ca.AGG_SUM_F64, 1)]
    cases = [
        [(0, ca.PRED_LT, 2500)],                          # sorted col: prunes
        [(0, ca.PRED_GE, 9000), (2, ca.PRED_GT, 0)],
        [(1, ca.PRED_GT, 10.0)],                          # NaN > c matches
        [(2, ca.PRED_EQ, 77)],                            # NULL-run chunks
        # OR group: a<500 OR a>9500 (same group id), AND b<=0
        [(0, ca.PRED_LT, 500, 7), (0, ca.PRED_GT, 9500, 7),
         (2, ca.PRED_LE, 0)],
    ]
    for preds in cases:
        res = {}
        for mode in ("1", "0"):
            _os.environ["CSTRIPE_DEVICE_PRUNE"] = mode
            try:
                with ca.Reader(path) as r, \
                     r.scan(cols_mask=0b111, preds=preds) as s:
                    filt = s.chunk_groups_filtered
                    s.stage()
                    parts = s.agg(aggs)
                res[mode] = (filt, [(p.i128, p.count, p.is_null,
                                     None if p.is_null else
                                     ("nan" if p.f64 != p.f64 else round(p.f64, 9)))
                                    for p in parts])
            finally:
                del _os.environ["CSTRIPE_DEVICE_PRUNE"]
        assert res["1"][0] == res["0"][0], preds    # identical chunk removal
        assert res["1"][1] == res["0"][1], preds
        if preds is cases[0]:
            assert res["1"][0] > 0                  # pruning actually fired


def test_zstd_canonical_with_nulls(tmp_path):
    """Sparse chunks over CANONICAL zstd streams: the value stream holds
    present rows only, so col_value's rank-mapped index must hit the same
    closed-form positions the oracle's decode sees — width-8 P, width-4
    flags, and NULL group keys together."""
    n = 40000
    q = ((RNG.integers(1, 51, n)) * 100).astype(np.int64)          # P(2)
    nq = (RNG.random(n) < 0.25).astype(np.uint8)
    fl = ca.text_slots(["A", "N", "R"][i % 3] for i in range(n))   # zr4b
    nf = (RNG.random(n) < 0.1).astype(np.uint8)
    path = str(tmp_path / "zn.cs")
    ca.write_table(path, [("q", ca.I64, 0), ("f", ca.TEXT, 0)],
                   [q, fl.view(np.int32)], nulls=[nq, nf],
                   compression=ca.COMP_ZSTD, chunk_group_row_limit=3000)
    aggs = [(ca.AGG_COUNT_COL, 0), (ca.AGG_SUM_I64, 0),
            (ca.AGG_MIN_I64, 0), (ca.AGG_MAX_I64, 0)]
    preds = [(0, ca.PRED_GT, 300)]
    op, ofilt, gp, gfilt = both(path, preds, aggs)
    assert ofilt == gfilt
    assert_parity(op, gp, aggs)
    # grouped by the nullable flag column: NULL key forms its own group
    with oracle.OracleTable(path) as t:
        og, _ = t.scan_agg([], aggs, group_cols=(1,))
    with ca.Reader(path) as r, r.scan(cols_mask=0b11, preds=[]) as s:
        s.stage()
        gg = s.agg_grouped(aggs, (1,))
    assert sorted(og.keys(), key=ca.encode_group_key) == \
           sorted(gg.keys(), key=ca.encode_group_key)
    assert len(og) == 4                       # A, N, R, NULL
    for k in og:
        for i in range(len(aggs)):
            assert og[k][i].i128 == gg[k][i].i128, (k, i)
            assert og[k][i].count == gg[k][i].count, (k, i)
