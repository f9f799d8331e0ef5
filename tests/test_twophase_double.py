"""Two-phase exchange breadth (VERDICT r1 item 6 slice): DOUBLE sums in the
YtStateRow format — meta bit1 marks a double state, the merge FP-adds
(udf/sum.c:27-35 double branch; front-query Merge
cg_fragment_compiler.cpp:4116-4134)."""
import ctypes as C

import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd._abi import YtStateRow, VT_INT64, VT_DOUBLE


def make_data(world, n=8000, keys=193):
    shards = []
    for r in range(world):
        rng = np.random.default_rng(900 + r)
        k = rng.integers(-keys // 2, keys // 2, n, dtype=np.int64)
        kn = (rng.random(n) < 0.02).astype(np.uint8)
        v = rng.random(n) * 100 - 50
        vn = (rng.random(n) < 0.08).astype(np.uint8)
        shards.append((k, kn, v, vn))
    return shards


def plan():
    return y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])


def chunk_of(shard):
    k, kn, v, vn = shard
    return y.Chunk([y.encode_int64(k, kn), y.encode_double(v, vn)], len(k))


def test_oracle_two_phase_double_sum():
    world = 3
    shards = make_data(world)
    # bottom queries: per-shard partials, partitioned by hash(key) % world
    parts = [[] for _ in range(world)]
    for r in range(world):
        states, counts = y.oracle_partial(plan(), chunk_of(shards[r]), world)
        at = 0
        for p in range(world):
            seg = (YtStateRow * max(counts[p], 1))()
            for i in range(counts[p]):
                seg[i] = states[at + i]
            parts[p].append((seg, counts[p]))
            at += counts[p]
    # front queries: per-partition merge; union must equal the single pass
    union = []
    for p in range(world):
        union += y.oracle_merge(plan(), parts[p])
    allk = np.concatenate([s[0] for s in shards])
    allkn = np.concatenate([s[1] for s in shards])
    allv = np.concatenate([s[2] for s in shards])
    allvn = np.concatenate([s[3] for s in shards])
    big = y.Chunk([y.encode_int64(allk, allkn), y.encode_double(allv, allvn)],
                  len(allk))
    want, _ = y.oracle_execute(plan(), big)
    wm = {r[0]: r for r in want}
    assert len(union) == len(want)
    for k, sv, cnt in union:
        _, wv, wc = wm[k]
        assert cnt == wc
        if wv is None:
            assert sv is None
        else:
            # double adds reassociate across the partition/merge order
            assert sv == pytest.approx(wv, rel=1e-9, abs=1e-9)


@pytest.mark.gpu
def test_gpu_two_phase_double_sum(cuda):
    shards = make_data(1, n=200_000, keys=4001)
    ch = chunk_of(shards[0])
    cap = 2 * 4001 + 1024
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, st = y.gpu_partial(plan(), ch.c_device(cuda), 1,
                               states_t.data_ptr(), cap,
                               max_groups_hint=4096)
    total = sum(counts)
    got, mst = y.gpu_merge(plan(), states_t.data_ptr(), total,
                           max_groups_hint=4096,
                           col_types=[VT_INT64, VT_DOUBLE])
    want, _ = y.oracle_execute(plan(), ch)
    gm = {r[0]: r for r in got}
    assert len(got) == len(want)
    for k, sv, cnt in want:
        gk, gs, gc = gm[k]
        assert gc == cnt
        if sv is None:
            assert gs is None
        else:
            assert gs == pytest.approx(sv, rel=1e-9, abs=1e-9)
