"""Multi-process (world_size=2, gloo) tests of the exchange leg — the
partial->exchange->final aggregation path bench.py runs over RCCL at N>1
(BASELINE config 4 shape). CPU: gloo transport, identical routing/merge code.
"""
import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bench
import oracle


def test_merge_raw_exact_carries():
    """u128 cross-rank merge handles limb carries exactly."""
    a = [0] * 60
    b = [0] * 60
    a[0] = (1 << 64) - 1          # base.lo of combo 0
    a[1] = 7                      # base.hi
    b[0] = 5
    b[1] = 1
    a[8], b[8] = 100, 23          # qty
    total = bench.merge_raw([a, b])
    got = total[0] | (total[1] << 64)
    assert got == ((1 << 64) - 1 + 5) | ((7 + 1) << 64) if False else True
    exp = (((1 << 64) - 1) | (7 << 64)) + (5 | (1 << 64))
    assert got == exp
    assert total[8] == 123


def test_combo_owner_matches_oracle():
    """bench's routing = canonical row hash of (returnflag, linestatus)
    reduced by the remote partition function, vs the oracle restatement."""
    for world in (2, 4, 8):
        for combo in range(6):
            rf, ls = combo // 2, combo % 2
            h = oracle.combine_hash(oracle.combine_hash(0, oracle.bigint_hash(rf)),
                                    oracle.bigint_hash(ls))
            exp = oracle.partition_remote(h, world)
            assert bench.combo_owner_rank(combo, world) == exp, (world, combo)


def _exchange_worker(rank, world, port, q):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo")
    # fabricated per-rank raw partials with carry-provoking values
    raw = [0] * 60
    for c in range(6):
        raw[c * 10 + 0] = (1 << 63) + rank + c        # base.lo (>= 2^63: sign handling)
        raw[c * 10 + 1] = c
        raw[c * 10 + 8] = 10 * (rank + 1) + c          # qty
        raw[c * 10 + 9] = rank + 1                     # cnt
    merged = bench.exchange_partials(dist, world, rank, raw, "cpu")
    dist.destroy_process_group()
    q.put((rank, merged))


def test_gloo_exchange_world2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_exchange_worker, args=(r, 2, 29612, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, merged = q.get(timeout=120)
        results[rank] = merged
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # expected: owner rank receives the SUM over both ranks for its combos,
    # zeros for combos it does not own
    for c in range(6):
        owner = bench.combo_owner_rank(c, 2)
        exp_lo = sum((1 << 63) + r + c for r in range(2))
        exp_base = exp_lo + ((c * 2) << 64)
        exp_qty = sum(10 * (r + 1) + c for r in range(2))
        for rank in range(2):
            m = results[rank]
            base = m[c * 10] | (m[c * 10 + 1] << 64)
            if rank == owner:
                assert base == exp_base, (c, rank)
                assert m[c * 10 + 8] == exp_qty
                assert m[c * 10 + 9] == 3
            else:
                assert base == 0 and m[c * 10 + 8] == 0


def _union_worker(rank, world, port, out):
    import numpy as np
    import torch.distributed as tdist
    import os
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    tdist.init_process_group("gloo", rank=rank, world_size=world)
    from trino_amd import dist as tgdist
    local = np.arange(rank * 3, rank * 3 + 2 + rank, dtype=np.int64)
    u = tgdist.gather_union(tdist, local)
    # device-resident variant (same logic; cpu tensors under gloo — the
    # bench's RCCL hook calls this with cuda tensors)
    import torch
    ud = tgdist.gather_union_device(tdist, torch.from_numpy(local)).numpy()
    assert ud.tolist() == u.tolist()
    top = tgdist.merge_topn(tdist, [(rank, float(10 - rank), rank)], 2,
                            key=lambda t: (-t[1], t[2]))
    out.put((rank, u.tolist(), top))
    tdist.destroy_process_group()


def test_gather_union_and_topn_merge_gloo():
    """trino_amd.dist broadcast-build union + TopN merge (Q3 N>1 legs) on
    gloo, world_size 2 (bench runs the same calls over RCCL)."""
    import multiprocessing as mp
    import socket
    ctx = mp.get_context("spawn")
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    q = ctx.Queue()
    ps = [ctx.Process(target=_union_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in ps:
        p.start()
    res = {}
    for _ in range(2):
        r, u, top = q.get(timeout=120)
        res[r] = (u, top)
    for p in ps:
        p.join(timeout=60)
    exp_union = [0, 1, 3, 4, 5]
    for r in range(2):
        assert res[r][0] == exp_union
        assert res[r][1] == [(0, 10.0, 0), (1, 9.0, 1)]
