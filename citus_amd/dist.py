"""
dist.py — the distributed combine step: per-GPU partial accumulators merged
across ranks, replacing the reference's coordinator merge of per-shard
partial rows over libpq (adaptive_executor.c:775-882 +
coordinator Agg built by MasterAggregateMutator,
multi_logical_optimizer.c:1831-1885).

Mechanism: ONE collective per query — torch.distributed.all_gather of the
fixed-size partial block (backend "nccl" IS RCCL over xGMI on ROCm; "gloo"
for the CPU-only multi-process tests) followed by the local cagg_combine,
which reproduces coord_combine_agg strictness (aggregate_utils.c:976-1000)
and COUNT's NULL->0 COALESCE on every rank. all_gather + local combine (not
an all_reduce) keeps the int128 fixed-point sums exact — carries cannot be
done inside a collective — and is byte-faithful to the reference shape:
the coordinator receives one partial row per shard and combines them.
Payload is O(100 B)/rank: latency-bound, far below xGMI link bandwidth.
"""
import ctypes as C

import torch
import torch.distributed as td

from . import Partial, combine as _combine

_PARTIAL_BYTES = C.sizeof(Partial)


def partials_to_tensor(parts, device="cpu"):
    buf = (Partial * len(parts))(*parts)
    t = torch.frombuffer(bytearray(C.string_at(buf, C.sizeof(buf))), dtype=torch.uint8)
    return t.to(device)


def tensor_to_partials(t, n):
    raw = bytes(t.cpu().numpy().tobytes())
    arr = (Partial * n).from_buffer_copy(raw)
    return list(arr)


def all_gather_combine(aggs, parts, device=None):
    """Gather every rank's partials (one collective) and combine locally.
    Returns the combined [Partial] (identical on every rank)."""
    world = td.get_world_size()
    if device is None:
        device = "cuda" if td.get_backend() == "nccl" else "cpu"
    mine = partials_to_tensor(parts, device)
    gathered = [torch.empty_like(mine) for _ in range(world)]
    td.all_gather(gathered, mine)
    parts_list = [tensor_to_partials(g, len(parts)) for g in gathered]
    return _combine(aggs, parts_list)
