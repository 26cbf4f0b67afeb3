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
