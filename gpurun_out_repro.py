import numpy as np, sys
sys.path.insert(0, "/root/repo")
import citus_amd as ca, oracle

def chk(name, cond):
    print(("OK " if cond else "FAIL ") + name, flush=True)

# 1. canonical P(2) dense
n = 25000
a = (np.arange(n, dtype=np.int64) * 7) % 5000
b = 1000 + (np.arange(n, dtype=np.int64) * 37) % 50000
path = "/tmp/c1.cs"
ca.write_table(path, [("a", ca.I64, 0), ("b", ca.I64, 0)], [a, b], compression=ca.COMP_LZ4)
print("wrote canonical", flush=True)
with ca.Reader(path) as r, r.scan(cols_mask=3, preds=[(0, ca.PRED_LT, 2400)]) as s:
    s.stage()
    print("staged", s.staged_bytes, flush=True)
    gp = s.agg([(ca.AGG_SUM_PROD_I64, 0, 1), (ca.AGG_COUNT_STAR, -1)])
    print("agg done", gp[0].i128, gp[1].count, flush=True)
    mask = a < 2400
    exp = int((a[mask].astype(object) * b[mask].astype(object)).sum())
    chk("q6ish", gp[0].i128 == exp and gp[1].count == int(mask.sum()))
# 2. next_batch canonical
with ca.Reader(path) as r, r.scan(cols_mask=3) as s:
    s.stage()
    v = np.zeros(10000, dtype=np.int64); e = np.zeros(10000, dtype=np.uint8)
    res = s.next_batch({0: v, 1: np.zeros(10000, dtype=np.int64)}, {0: e, 1: np.zeros(10000, dtype=np.uint8)})
    nr, _ = res
    chk("batch", bool((v[:nr] == a[:nr]).all()))
# 3. grouped with canonical measure + greedy i8 keys
k = (np.arange(n) % 3).astype(np.int8)
path2 = "/tmp/c2.cs"
ca.write_table(path2, [("k", ca.I8, 0), ("v", ca.I64, 0)], [k, b], compression=ca.COMP_LZ4)
with ca.Reader(path2) as r, r.scan(cols_mask=3) as s:
    s.stage()
    res = s.agg_grouped([(ca.AGG_SUM_I64, 1), (ca.AGG_COUNT_STAR, -1)], (0,))
    ok = all(res[(g,)][0].i128 == int(b[k == g].sum()) for g in (0, 1, 2))
    chk("grouped", ok)
# 4. sparse canonical
nulls = (np.arange(n) % 5 == 0).astype(np.uint8)
path3 = "/tmp/c3.cs"
ca.write_table(path3, [("a", ca.I64, 0)], [a], nulls=[nulls], compression=ca.COMP_LZ4)
with ca.Reader(path3) as r, r.scan(cols_mask=1) as s:
    s.stage()
    gp = s.agg([(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_COL, 0)])
    m2 = nulls == 0
    chk("sparse", gp[0].i128 == int(a[m2].sum()) and gp[1].count == int(m2.sum()))
# 5. const + f64 canonical
c = np.full(n, -42, dtype=np.int64)
f = (np.uint64(0x3FF0000000000000) | (np.arange(n, dtype=np.uint64) % 251)).view(np.float64)
path4 = "/tmp/c4.cs"
ca.write_table(path4, [("c", ca.I64, 0), ("f", ca.F64, 0)], [c, f], compression=ca.COMP_LZ4)
with ca.Reader(path4) as r, r.scan(cols_mask=3) as s:
    s.stage()
    gp = s.agg([(ca.AGG_SUM_I64, 0), (ca.AGG_SUM_F64, 1), (ca.AGG_MIN_F64, 1)])
    chk("constf64", gp[0].i128 == int(c.sum()) and abs(gp[1].f64 - f.sum()) < 1e-6 * abs(f.sum()))
print("ALL DONE", flush=True)
