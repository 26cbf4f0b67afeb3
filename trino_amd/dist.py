"""Distributed exchange helpers for the query pipelines.

Mirrors the reference's two join distribution modes at this path
(sql/planner/optimizations/DetermineJoinDistributionType.java):
 - BROADCAST (replicated build): the filtered customer keys are small
   (~1/5 of customers), so every rank gathers the union and builds the
   same hash table locally — `gather_union`;
 - the orders⨝lineitem leg is co-partitioned by construction (each rank
   generates lineitem for its own order range), so it needs no exchange.

Final ORDER BY ... LIMIT merges per-rank TopN results on rank 0
(`merge_topn`) — orderkey ranges are rank-disjoint, so per-rank group
results never need a cross-rank combine.

Backend-agnostic: `all_gather_object` works over both nccl (RCCL, bench)
and gloo (CPU tests, world_size 2).
"""
import numpy as np


def gather_union(tdist, arr):
    """All-gather variable-length numpy arrays; returns the rank-ordered
    concatenation (same result on every rank)."""
    world = tdist.get_world_size()
    objs = [None] * world
    tdist.all_gather_object(objs, np.ascontiguousarray(arr))
    return np.concatenate([np.asarray(o) for o in objs])


def merge_topn(tdist, rows, limit, key):
    """Gather per-rank TopN candidate rows and re-select the global TopN.
    `rows` is a list of tuples; `key` maps a tuple to its sort key
    (ascending). Returns the merged list on every rank."""
    world = tdist.get_world_size()
    objs = [None] * world
    tdist.all_gather_object(objs, list(rows))
    merged = [t for o in objs for t in o]
    merged.sort(key=key)
    return merged[:limit]


def gather_union_device(tdist, t):
    """Device-resident variable-length all-gather (RCCL over xGMI for cuda
    tensors; same code path runs on cpu tensors under gloo for the
    world-size-2 CPU tests): lengths all-gather, pad to max, tensor
    all-gather, trim + concat. No host round-trip."""
    import torch
    world = tdist.get_world_size()
    n = torch.tensor([t.numel()], dtype=torch.int64, device=t.device)
    ns = [torch.zeros_like(n) for _ in range(world)]
    tdist.all_gather(ns, n)
    counts = [int(x.item()) for x in ns]
    m = max(counts + [1])
    pad = torch.zeros(m, dtype=t.dtype, device=t.device)
    if t.numel():
        pad[:t.numel()] = t
    outs = [torch.empty(m, dtype=t.dtype, device=t.device) for _ in range(world)]
    tdist.all_gather(outs, pad)
    return torch.cat([o[:c] for o, c in zip(outs, counts)])
