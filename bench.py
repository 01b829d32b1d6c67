#!/usr/bin/env python3
"""
bench.py — the BASELINE.json headline: TPC-H Q6 filter+SUM over synthetic
lineitem-shaped LZ4 columnar stripes, rows/s + achieved HBM GB/s on MI355X.

A step = one pass of the hot path over the staged stripes:
  host chunk-group pruning -> LZ4 segment decode kernel -> fused
  filter+partial-aggregate kernel -> device-wide reduce -> combine
  (all_gather over RCCL at N>1 — the coordinator-merge replacement).
Data is device-resident when the timed region starts (staged HtoD untimed);
decode+filter+aggregate re-run every step — no work is skipped.

N=1 workload is BASELINE config 2 (the largest single-GPU config): 100M-row
lineitem, lz4, stripe 150k / chunk group 10k (reference defaults). N>1:
per-rank shards of the same size (weak scaling; config-3 shape).

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--rows R]
Multi-GPU via: python -m torch.distributed.run --nnodes=1 --nproc-per-node N
               --master-addr 127.0.0.1 bench.py --gpus N ...
"""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

# algorithmic bytes/row = decompressed projected column widths + exists bits,
# counted once
ALG_BYTES = {"q6": 32.5,       # 4 int64 + 4 exists bits
             "q1": 48.875,     # 5 int64 + 2 varlena char(1) slots + 7 exists bits
             "count": 8.125}   # 1 int64 + 1 exists bit
HBM_PEAK_GBPS = 8000.0     # MI355X HBM3E spec peak (MI355X_MICROARCH.md)


def build_if_needed():
    if not os.path.exists(os.path.join(REPO, "citus_amd", "libcstripe.so")):
        import subprocess
        subprocess.check_call(["make", "-C", os.path.join(REPO, "citus_amd", "csrc")])
    if not os.path.exists(os.path.join(REPO, "oracle", "liboracle.so")):
        import subprocess
        subprocess.check_call(["make", "-C", os.path.join(REPO, "oracle")])


def q6_preds(ca):
    # l_shipdate >= 1994-01-01 (day 8766) AND < 1995-01-01 (9131)
    # AND l_discount BETWEEN 0.05 AND 0.07 AND l_quantity < 24
    return [(5, ca.PRED_GE, 8766), (5, ca.PRED_LT, 9131),
            (3, ca.PRED_GE, 5), (3, ca.PRED_LE, 7), (1, ca.PRED_LT, 2400)]


def q6_aggs(ca):
    return [(ca.AGG_SUM_PROD_I64, 2, 3), (ca.AGG_COUNT_STAR, -1)]


def q1_preds(ca):
    return [(5, ca.PRED_LE, 10471)]   # l_shipdate <= 1998-09-02


def q1_aggs(ca):
    return [(ca.AGG_SUM_I64, 1), (ca.AGG_SUM_I64, 2),
            (ca.AGG_SUM_DISC_I64, 2, 3, -1, 100),
            (ca.AGG_SUM_DISC_TAX_I64, 2, 3, 4, 100),
            (ca.AGG_COUNT_STAR, -1)]


def count_preds(ca):
    return [(1, ca.PRED_LT, 2400)]    # l_quantity < 24 (config 1)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=0,
                    help="timed steps (0 = auto: ~10 s of steady-state steps)")
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--rows", type=int, default=1_000_000_000,
                    help="rows per GPU (default 1B: the north-star single-GPU "
                         "config; config 2 = 100M)")
    ap.add_argument("--compression", default="lz4", choices=["lz4", "none", "zstd"])
    ap.add_argument("--query", default="q6", choices=["q6", "q1", "count"])
    ap.add_argument("--seg-bytes", type=int, default=0,
                    help="LZ4 micro-segment size override (bytes)")
    ap.add_argument("--min-match", type=int, default=0,
                    help="writer LZ4 min match length (>=4; GPU-decode knob)")
    ap.add_argument("--canonical", type=int, default=1,
                    help="writer canonical stream modes (closed-form GPU "
                         "access); 0 = greedy parse (round-1 behaviour)")
    ap.add_argument("--shards", type=int, default=0,
                    help="shard files per GPU scanned as one table (config 3: "
                         "4 shards/GPU x 8 GPUs = 32 shards). 0 = auto: 16 for "
                         ">=100M rows (parallel generation), else 1")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    args = ap.parse_args()

    build_if_needed()
    import citus_amd as ca

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = world > 1

    import torch
    # initialize torch's device view BEFORE libcstripe touches HIP (the two
    # ROCm runtime instances coexist only in that order; tests/conftest.py)
    torch.cuda.is_available()
    rccl_comm = None
    if dist:
        import torch.distributed as td
        from citus_amd.dist import make_rccl_comm
        torch.cuda.set_device(local_rank)
        td.init_process_group("nccl")
        # the combine data path runs through the C ABI (cagg_combine_rccl,
        # librccl); torch.distributed only bootstraps the unique id and
        # handles the timing all-reduce
        rccl_comm = make_rccl_comm(local_rank)

    comp = {"lz4": ca.COMP_LZ4, "none": ca.COMP_NONE, "zstd": ca.COMP_ZSTD}[args.compression]

    # ---- setup (untimed): generate per-rank shards, open, prune, stage ----
    if args.shards == 0:
        args.shards = 32 if args.rows >= 100_000_000 else 1
    base_seed = 42 + rank * args.shards
    # default to tmpfs: the 1B dataset is ~21 GB/rank and the 8-GPU scale run
    # needs 8 of them — /dev/shm is host-RAM-sized, container /tmp often isn't
    default_cache = ("/dev/shm/cstripe_bench" if os.path.isdir("/dev/shm")
                     else "/tmp/cstripe_bench")
    cache = os.environ.get("CSTRIPE_BENCH_DIR", default_cache)
    os.makedirs(cache, exist_ok=True)
    seg_kb = -args.seg_bytes if args.seg_bytes else 0
    if args.shards > 1:
        shard = os.path.join(cache, f"li_{args.rows}_{args.compression}_s{args.seg_bytes}"
                                    f"_m{args.min_match}_c{args.canonical}_n{args.shards}_r{rank}")
        t0 = time.time()
        if not os.path.exists(os.path.join(shard, f"shard{args.shards - 1:02d}.cs")):
            # parallel generation, one writer thread per shard; when several
            # ranks generate on one host, split the cores between them
            ncpu = os.cpu_count() or 8
            ca.gen_lineitem_shards(shard, args.rows, args.shards,
                                   base_seed=base_seed, compression=comp,
                                   seg_kb=seg_kb, min_match=args.min_match,
                                   canonical=args.canonical,
                                   threads=max(2, ncpu // max(1, world)))
    else:
        shard = os.path.join(cache, f"li_{args.rows}_{args.compression}_s{args.seg_bytes}"
                                    f"_m{args.min_match}_c{args.canonical}_r{rank}.cs")
        t0 = time.time()
        if not os.path.exists(shard):
            ca.gen_lineitem(shard, args.rows, seed=base_seed, compression=comp,
                            seg_bytes=args.seg_bytes, min_match=args.min_match,
                            canonical=args.canonical)
    gen_s = time.time() - t0

    reader = ca.Reader(shard)
    if args.query == "q6":
        preds, aggs, group_cols = q6_preds(ca), q6_aggs(ca), None
    elif args.query == "q1":
        preds, aggs, group_cols = q1_preds(ca), q1_aggs(ca), (6, 7)
    else:
        preds, aggs, group_cols = count_preds(ca), [(ca.AGG_COUNT_STAR, -1)], None
    mask = ca.agg_cols_mask(aggs)
    if group_cols:
        for c in group_cols:
            mask |= 1 << c
    scan = reader.scan(cols_mask=mask, preds=preds)
    t0 = time.time()
    scan.stage(local_rank if dist else -1)
    stage_s = time.time() - t0
    staged_gb = scan.staged_bytes / 1e9

    def step():
        if group_cols:
            res = scan.agg_grouped(aggs, group_cols)
            if dist:
                from citus_amd.dist import all_gather_combine_grouped_rccl
                return all_gather_combine_grouped_rccl(rccl_comm, aggs, res)
            return {k: ca.combine(aggs, [parts]) for k, parts in res.items()}
        parts = scan.agg(aggs)
        if dist:
            return rccl_comm.combine(aggs, parts)
        return ca.combine(aggs, [parts])

    # ---- warmup (also calibrates auto step count) ----
    t0 = time.time()
    for _ in range(max(1, args.warmup)):
        result = step()
    per_step_est = (time.time() - t0) / max(1, args.warmup)
    if args.steps == 0:
        # ~20 s of timed steps: long enough for the driver-side activity
        # sampler to observe the GPU busy, bounded for tiny workloads
        args.steps = max(10, min(40000, int(20.0 / max(per_step_est, 1e-4))))
        if dist:
            # every rank MUST run the same step count (the per-step combine
            # is a collective): take the max of the per-rank calibrations
            t = torch.tensor([args.steps], device="cuda")
            td.all_reduce(t, op=td.ReduceOp.MAX)
            args.steps = int(t.item())

    decode_ms = []
    agg_ms = []

    if dist:
        import torch.distributed as td
        td.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t_start = time.time()
    for _ in range(args.steps):
        result = step()
        decode_ms.append(scan.last_decode_ms)
        agg_ms.append(scan.last_agg_ms)
    if dist:
        td.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    elapsed = time.time() - t_start

    if dist:
        t = torch.tensor([elapsed], device="cuda")
        td.all_reduce(t, op=td.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank != 0:
        scan.end()
        reader.close()
        if dist:
            rccl_comm.close()
            td.destroy_process_group()
        return

    total_rows = args.rows * world
    value = total_rows * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    alg_bytes_per_row = ALG_BYTES[args.query]
    avg_decode = sum(decode_ms) / len(decode_ms)
    avg_agg = sum(agg_ms) / len(agg_ms)
    if scan.last_fused:
        dominant = "fused_agg_kernel" if not group_cols else "fused_grouped_kernel"
    elif avg_decode >= avg_agg:
        dominant = ("zr_decode_lds2_kernel" if args.compression == "zstd"
                    else "lz4_decode_lane_kernel")
    else:
        # all-dense scans (canonical writer default) run the multi-row kernels
        dominant = "multi_grouped_kernel" if group_cols else "multi_agg_kernel"
    dominant_ms = max(avg_decode, avg_agg)
    alg_bytes = args.rows * alg_bytes_per_row          # per launch (this rank)
    achieved_gbps = alg_bytes / (dominant_ms / 1e3) / 1e9 if dominant_ms > 0 else None

    # PMC-measured HBM traffic for the dominant kernel, if a committed profile
    # exists (profiles/traffic_r*.json, written from rocprofv3 --pmc runs)
    traffic = None
    try:
        import glob
        profs = sorted(glob.glob(os.path.join(REPO, "profiles", "traffic_r*.json")))
        if profs:
            tj = json.load(open(profs[-1]))
            if (tj.get("workload_rows") == args.rows and tj.get("kernel") == dominant
                    and tj.get("workload") == f"{args.query}_lineitem_{args.rows // 1_000_000}m_{args.compression}"):
                traffic = tj.get("hbm_bytes_per_launch")
    except Exception:
        traffic = None

    # ---- in-run parity pin: generator-exact expected values (any size) ----
    # csbench_expected_q6/_q1 recompute the query straight from the seeded
    # generator streams with int128 on host CPU (bench_gen.cpp) — an
    # independent end-to-end check of writer -> stage -> decode -> filter ->
    # aggregate -> combine at the FULL benchmark size.
    parity = None
    if rank == 0:
        # expected values cover EVERY rank's shards (seeds are deterministic:
        # rank r uses base 42 + r*shards), so the pin also holds for the
        # multi-GPU combined result
        shard_rows = [args.rows // args.shards + (args.rows % args.shards if i == 0 else 0)
                      for i in range(args.shards)]
        all_seeds = [42 + r * args.shards + i
                     for r in range(world) for i in range(args.shards)]
        all_rows = shard_rows * world
        if args.query == "q6":
            exp_rev, exp_cnt = 0, 0
            for nrows, sd in zip(all_rows, all_seeds):
                r_, c_ = ca.expected_q6(nrows, sd)
                exp_rev += r_
                exp_cnt += c_
            parity = ("bit-exact" if (result[0].i128 == exp_rev and
                                      result[1].count == exp_cnt)
                      else "MISMATCH")
        elif args.query == "q1":
            # generator flags are real char(1) varlena columns: group keys
            # are the payload bytes; expected_q1 indexes map to those chars
            RF = {0: ord("A"), 1: ord("N"), 2: ord("R")}
            LS = {0: ord("O"), 1: ord("F")}
            exp = {}
            for nrows, sd in zip(all_rows, all_seeds):
                for k, v in ca.expected_q1(nrows, sd).items():
                    kk = (RF[k[0]], LS[k[1]])
                    cur = exp.setdefault(kk, [0, 0, 0, 0, 0])
                    for j in range(5):
                        cur[j] += v[j]
            exp = {k: v for k, v in exp.items() if v[4] > 0}
            ok = set(exp) == set(result)
            if ok:
                for k, v in exp.items():
                    got = result[k]
                    ok = ok and [got[0].i128, got[1].i128, got[2].i128,
                                 got[3].i128, got[4].count] == v
            parity = "bit-exact" if ok else "MISMATCH"

    # ---- CPU baseline (oracle restatement, rank 0, N=1 only) ----
    cpu_baseline = None
    cpu_mt = None
    if world == 1 and not args.no_cpu_baseline:
        import oracle
        # bounded sample: ONE shard file of the same dataset (~10-30 s CPU)
        if args.shards > 1:
            sample_path = os.path.join(shard, "shard01.cs")
            sample_rows = args.rows // args.shards
            sample_desc = f"one {sample_rows / 1e6:.0f}M-row shard of the same dataset"
        else:
            sample_path = shard
            sample_rows = args.rows
            sample_desc = f"full {sample_rows / 1e6:.0f}M-row pass"
        with oracle.OracleTable(sample_path) as t:
            t0 = time.time()
            if group_cols:
                t.scan_agg(preds, aggs, group_cols=group_cols)
            else:
                t.scan_agg(preds, aggs)
            cpu_s = time.time() - t0
            cpu_mt = None
            if not group_cols:
                t0 = time.time()
                _mt_parts, mt_cores = t.scan_agg_mt(preds, aggs)
                cpu_mt = {"value": sample_rows / (time.time() - t0),
                          "unit": "rows/s", "cores": mt_cores, "kind": "port",
                          "sample": f"all-core, {sample_desc}"}
        cpu_baseline = {
            "value": sample_rows / cpu_s,
            "unit": "rows/s",
            "cores": 1,
            "kind": "port",
            "sample": f"{sample_desc}, single thread ({cpu_s:.1f}s)",
        }

    out = {
        "metric": f"columnar_rows_per_s_{args.query}",
        "value": value,
        "unit": "rows/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": ms_per_step,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "int64",
        "data": "synthetic",
        "config": {
            "workload": f"{args.query}_lineitem_{args.rows // 1_000_000}m_{args.compression}",
            "query": f"tpch_{args.query}" if args.query != "count" else "count_star",
            "rows_per_gpu": args.rows,
            "compression": args.compression,
            "stripe_rows": 150000,
            "chunk_group_rows": 10000,
            "parallelism": f"shard-dp{world}x{args.shards}",
            "staged_gb_compressed": round(staged_gb, 3),
            "gen_s": round(gen_s, 1),
            "stage_s": round(stage_s, 2),
            "kernel_ms": {"decode": round(avg_decode, 3), "filter_agg": round(avg_agg, 3)},
            "parity_vs_oracle": parity,
            "q6_revenue_scale4": (result[0].i128 if world == 1 and args.query == "q6"
                                  else None),
            "n_groups": len(result) if group_cols else None,
        },
        "roofline": {
            "bound": "hbm",
            "achieved": round(achieved_gbps, 1) if achieved_gbps else None,
            "peak": HBM_PEAK_GBPS,
            "unit": "GB/s",
            "frac": round(achieved_gbps / HBM_PEAK_GBPS, 4) if achieved_gbps else None,
            "traffic": traffic,
            "kernel": dominant,
        },
        "cpu_baseline": cpu_baseline,
        "cpu_baseline_allcore": cpu_mt if (world == 1 and not args.no_cpu_baseline) else None,
    }
    print(json.dumps(out))

    scan.end()
    reader.close()
    if dist:
        rccl_comm.close()
        td.destroy_process_group()


if __name__ == "__main__":
    main()
