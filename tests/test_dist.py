"""
Multi-process combine path (gloo, world_size 2, CPU): each rank scans its own
shard through the ORACLE (CPU stand-in for the per-GPU partial — the GPU
variant is covered in test_gpu_parity.py), then the product combine path
(citus_amd.dist.all_gather_combine -> cagg_combine) merges across ranks.
The result must equal a single-scan oracle run over the union — exactly the
reference contract: one partial row per shard, combined once at the
coordinator.
"""
import json
import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json, os, sys
sys.path.insert(0, os.environ["REPO"])
import torch.distributed as td
import citus_amd as ca
import oracle
from citus_amd.dist import all_gather_combine

td.init_process_group("gloo")
rank = td.get_rank()
shard = os.path.join(os.environ["SHARD_DIR"], f"shard{rank}.cs")
preds = [(1, ca.PRED_LT, 2400)]
aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_PROD_I64, 2, 3),
        (ca.AGG_MIN_I64, 1), (ca.AGG_MAX_I64, 2), (ca.AGG_SUM_I64, 1)]
with oracle.OracleTable(shard) as t:
    parts, _ = t.scan_agg(preds, aggs)
combined = all_gather_combine(aggs, parts, device="cpu")
if rank == 0:
    print("RESULT " + json.dumps([p.as_dict() for p in combined]))
td.destroy_process_group()
"""


@pytest.mark.timeout(300)
def test_two_rank_combine_matches_single_scan(tmp_path):
    import citus_amd as ca
    import oracle

    # two shards: hash-partitioned lineitem-shaped data (one file per shard,
    # the static 'shard group per rank' mapping of SURVEY.md §8e)
    ca.gen_lineitem(str(tmp_path / "shard0.cs"), 60_000, seed=42)
    ca.gen_lineitem(str(tmp_path / "shard1.cs"), 60_000, seed=4242)
    # union file for the single-scan answer
    # (scan both shards separately through the oracle and combine host-side)
    preds = [(1, ca.PRED_LT, 2400)]
    aggs = [(ca.AGG_COUNT_STAR, -1), (ca.AGG_SUM_PROD_I64, 2, 3),
            (ca.AGG_MIN_I64, 1), (ca.AGG_MAX_I64, 2), (ca.AGG_SUM_I64, 1)]
    partials = []
    for s in (0, 1):
        with oracle.OracleTable(str(tmp_path / f"shard{s}.cs")) as t:
            p, _ = t.scan_agg(preds, aggs)
            partials.append(p)
    expect = [p.as_dict() for p in ca.combine(aggs, partials)]

    env = dict(os.environ, REPO=REPO, SHARD_DIR=str(tmp_path),
               MASTER_ADDR="127.0.0.1", MASTER_PORT="29517")
    script = str(tmp_path / "worker.py")
    open(script, "w").write(WORKER)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", script],
        env=env, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")]
    assert line, out.stdout
    got = json.loads(line[0][len("RESULT "):])
    assert got == expect


WORKER_GROUPED = r"""
import json, os, sys
sys.path.insert(0, os.environ["REPO"])
import torch.distributed as td
import citus_amd as ca
import oracle
from citus_amd.dist import all_gather_combine_grouped

td.init_process_group("gloo")
rank = td.get_rank()
shard = os.path.join(os.environ["SHARD_DIR"], f"shard{rank}.cs")
preds = [(5, ca.PRED_LE, 10471)]
aggs = [(ca.AGG_SUM_I64, 1), (ca.AGG_SUM_DISC_I64, 2, 3, -1, 100),
        (ca.AGG_COUNT_STAR, -1)]
with oracle.OracleTable(shard) as t:
    res, _ = t.scan_agg(preds, aggs, group_cols=(6, 7))
combined = all_gather_combine_grouped(aggs, res, device="cpu")
if rank == 0:
    out = {f"{k[0]},{k[1]}": [p.as_dict() for p in v] for k, v in combined.items()}
    print("RESULT " + json.dumps(out))
td.destroy_process_group()
"""


@pytest.mark.timeout(300)
def test_two_rank_grouped_combine(tmp_path):
    import citus_amd as ca
    import oracle

    ca.gen_lineitem(str(tmp_path / "shard0.cs"), 50_000, seed=42)
    ca.gen_lineitem(str(tmp_path / "shard1.cs"), 50_000, seed=4242)
    preds = [(5, ca.PRED_LE, 10471)]
    aggs = [(ca.AGG_SUM_I64, 1), (ca.AGG_SUM_DISC_I64, 2, 3, -1, 100),
            (ca.AGG_COUNT_STAR, -1)]
    tables = []
    for s in (0, 1):
        with oracle.OracleTable(str(tmp_path / f"shard{s}.cs")) as t:
            res, _ = t.scan_agg(preds, aggs, group_cols=(6, 7))
            tables.append(res)
    keys = sorted(set(tables[0]) | set(tables[1]))
    null_row = [ca.Partial(is_null=1) for _ in aggs]
    expect = {f"{k[0]},{k[1]}": [p.as_dict() for p in
                                 ca.combine(aggs, [t.get(k, null_row) for t in tables])]
              for k in keys}

    env = dict(os.environ, REPO=REPO, SHARD_DIR=str(tmp_path),
               MASTER_ADDR="127.0.0.1", MASTER_PORT="29531")
    script = str(tmp_path / "workerg.py")
    open(script, "w").write(WORKER_GROUPED)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", script],
        env=env, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")]
    got = json.loads(line[0][len("RESULT "):])
    assert got == expect


@pytest.mark.gpu
def test_rccl_combine_single_rank():
    """The C-ABI RCCL combine path executed on hardware: comm init, the
    ncclAllGather byte round trip, and cagg_combine_rccl semantics at
    world size 1 (the 8-GPU run is the driver's; 2-rank semantics are
    covered by the gloo tests above)."""
    import citus_amd as ca
    assert ca.gpu_available()
    uid = ca.RcclComm.unique_id()
    with ca.RcclComm(1, 0, uid, device=0) as comm:
        blob = b"cstripe-rccl-roundtrip" * 11
        assert comm.allgather_bytes(blob) == blob
        aggs = [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1), (ca.AGG_MIN_I64, 0)]
        parts = [ca.Partial(i128_lo=-5, i128_hi=-1, count=3),
                 ca.Partial(i128_lo=3, count=3),
                 ca.Partial(i128_lo=-7, i128_hi=-1, count=3)]
        out = comm.combine(aggs, parts)
        assert out[0].i128 == -5
        assert out[1].count == 3
        assert out[2].i128 == -7


@pytest.mark.gpu
def test_rccl_grouped_combine_single_rank():
    import citus_amd as ca
    from citus_amd import dist as cdist
    uid = ca.RcclComm.unique_id()
    with ca.RcclComm(1, 0, uid, device=0) as comm:
        aggs = [(ca.AGG_SUM_I64, 0), (ca.AGG_COUNT_STAR, -1)]
        res = {(0, 1): [ca.Partial(i128_lo=10, count=2), ca.Partial(i128_lo=2, count=2)],
               (2, 0): [ca.Partial(i128_lo=-4, i128_hi=-1, count=1),
                        ca.Partial(i128_lo=1, count=1)]}
        out = cdist.all_gather_combine_grouped_rccl(comm, aggs, res)
        assert set(out) == {(0, 1), (2, 0)}
        assert out[(0, 1)][0].i128 == 10
        assert out[(2, 0)][0].i128 == -4
