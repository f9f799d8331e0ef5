"""Multi-process CPU coverage of the 8-GPU exchange logic:
world_size-2 gloo all-to-all of YtStateRow partitions + merge, against a
single-pass oracle run on the concatenated data. On GPUs the same layout goes
through RCCL (`nccl` backend) over xGMI — bench.py's --gpus N path."""
import ctypes as C
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import ytsaurus_amd as y
from ytsaurus_amd._abi import YtStateRow

N_PER_RANK = 20000
KEYS = 257


def make_shard(rank):
    rng = np.random.default_rng(100 + rank)
    keys = rng.integers(-KEYS // 2, KEYS // 2, N_PER_RANK, dtype=np.int64)
    vals = rng.integers(-10**6, 10**6, N_PER_RANK, dtype=np.int64)
    knull = (rng.random(N_PER_RANK) < 0.01).astype(np.uint8)
    vnull = (rng.random(N_PER_RANK) < 0.05).astype(np.uint8)
    return keys, vals, knull, vnull


def make_plan():
    return y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])


def _worker(rank, world, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29781"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        keys, vals, knull, vnull = make_shard(rank)
        chunk = y.Chunk([y.encode_int64(keys, knull), y.encode_int64(vals, vnull)],
                        N_PER_RANK)
        plan = make_plan()
        states, counts = y.oracle_partial(plan, chunk, world)

        # pack each partition's states into an int64 tensor [n, 4]
        send = []
        at = 0
        for p in range(world):
            m = np.zeros((counts[p], 4), dtype=np.uint64)
            for i in range(counts[p]):
                s = states[at + i]
                m[i] = (s.key_bits, s.meta, s.sum_bits, s.row_count)
            send.append(torch.from_numpy(m.view(np.int64)))
            at += counts[p]

        # exchange partitioned state rows. gloo has no all-to-all, so the CPU
        # coverage uses all_gather + local selection — same partition/merge
        # logic; the GPU path does RCCL all_to_all_single (bench.py).
        gathered = [None] * world
        dist.all_gather_object(gathered, [t.numpy() for t in send])
        recv = [torch.from_numpy(gathered[src][rank].copy()) for src in range(world)]

        # local merge of this rank's partition
        mine = torch.cat(recv, dim=0) if recv else torch.zeros((0, 4), dtype=torch.int64)
        mu = mine.numpy().view(np.uint64)
        arr = (YtStateRow * max(len(mine), 1))()
        for i in range(len(mine)):
            arr[i] = YtStateRow(key_bits=int(mu[i][0]), meta=int(mu[i][1]),
                                sum_bits=int(mu[i][2]), row_count=int(mu[i][3]))
        merged = y.oracle_merge(make_plan(), [(arr, len(mine))])
        q.put((rank, merged))
    finally:
        dist.destroy_process_group()


def test_gloo_two_phase_matches_single_pass():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, merged = q.get(timeout=300)
        results[rank] = merged
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    union = [r for rank in results for r in results[rank]]

    # single-pass reference on the concatenated data
    allk = np.concatenate([make_shard(r)[0] for r in range(world)])
    allv = np.concatenate([make_shard(r)[1] for r in range(world)])
    allkn = np.concatenate([make_shard(r)[2] for r in range(world)])
    allvn = np.concatenate([make_shard(r)[3] for r in range(world)])
    chunk = y.Chunk([y.encode_int64(allk, allkn), y.encode_int64(allv, allvn)],
                    len(allk))
    want, _ = y.oracle_execute(make_plan(), chunk)
    assert y.sort_rows(union) == y.sort_rows(want)

    # partitions are disjoint by key
    keys_by_rank = [set(r[0] for r in results[rank]) for rank in results]
    assert not (keys_by_rank[0] & keys_by_rank[1])
