"""Perf probe: gen N-row lineitem, stage, run Q6 scan_agg 3x and print
per-pass kernel ms (used for on-hardware A/B iterations this round)."""
import sys, time
sys.path.insert(0, "/root/repo")
import citus_amd as ca

rows = int(sys.argv[1]) if len(sys.argv) > 1 else 30_000_000
path = f"/tmp/li_{rows}.cs"
t0 = time.time()
ca.gen_lineitem(path, rows, seed=42)
print(f"gen {rows} rows in {time.time()-t0:.1f}s", flush=True)
preds = [(5, ca.PRED_GE, 8766), (5, ca.PRED_LT, 9131),
         (3, ca.PRED_GE, 5), (3, ca.PRED_LE, 7), (1, ca.PRED_LT, 2400)]
aggs = [(ca.AGG_SUM_PROD_I64, 2, 3), (ca.AGG_COUNT_STAR, -1)]
with ca.Reader(path) as r, r.scan(cols_mask=ca.agg_cols_mask(aggs), preds=preds) as s:
    t0 = time.time()
    s.stage()
    print(f"staged {s.staged_bytes/1e9:.3f} GB in {time.time()-t0:.1f}s", flush=True)
    for it in range(3):
        t0 = time.time()
        gp = s.agg(aggs)
        print(f"agg[{it}] revenue={gp[0].i128} count={gp[1].count} "
              f"kms={s.last_kernel_ms:.3f} fused={s.last_fused} wall={time.time()-t0:.2f}s", flush=True)
print("DONE", rows, flush=True)
