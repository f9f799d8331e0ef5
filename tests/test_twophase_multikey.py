"""Multi-key two-phase exchange: composite keys packed
with CROSS-RANK-COMMON zigzag ranges (yt_gpu_key_ranges -> caller min/max
reduce -> partial_mk/merge_mk), so every rank's packed key_bits agree and
the 32-byte YtStateRow format + merge kernels carry multi-key plans
unchanged. Mirrors the reference's key-prefix shuffle partition
(shuffling_reader.cpp:40-42, cg_fragment_compiler.cpp:4016-4134)."""
import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd._abi import VT_INT64, VT_BOOLEAN


def make_shards(world, n=6000):
    shards = []
    for r in range(world):
        rng = np.random.default_rng(1200 + r)
        k0 = rng.integers(-40 - 10 * r, 40, n, dtype=np.int64)  # rank-varying range
        k0n = (rng.random(n) < 0.03).astype(np.uint8)
        k1 = rng.integers(0, 7, n, dtype=np.int64)
        v = rng.integers(-10**6, 10**6, n, dtype=np.int64)
        vn = (rng.random(n) < 0.05).astype(np.uint8)
        shards.append((k0, k0n, k1, v, vn))
    return shards


def chunk_of(s):
    k0, k0n, k1, v, vn = s
    return y.Chunk([y.encode_int64(k0, k0n), y.encode_int64(k1),
                    y.encode_int64(v, vn)], len(k0))


def plan():
    return y.Plan(keys=[y.col(0), y.col(1)],
                  aggs=[y.agg_sum(y.col(2)), y.agg_sum1()])


def reduce_ranges(per_rank):
    k = len(per_rank[0][0])
    zzmin = [min(pr[0][i] for pr in per_rank) for i in range(k)]
    zzmax = [max(pr[1][i] for pr in per_rank) for i in range(k)]
    return zzmin, zzmax


def oracle_key_ranges(shard):
    # zigzag-space bounds per key column, host-side (tests only)
    def zz(a):
        a = np.asarray(a, dtype=np.int64)
        return ((a.astype(np.uint64) << np.uint64(1))
                ^ (a >> np.int64(63)).astype(np.uint64))
    k0, k0n, k1, _, _ = shard
    out_min, out_max = [], []
    for col, nul in ((k0, k0n), (k1, None)):
        z = zz(col)
        if nul is not None and nul.any():
            z = z[nul == 0]
        out_min.append(int(z.min()))
        out_max.append(int(z.max()))
    return out_min, out_max


def test_oracle_multikey_two_phase():
    world = 3
    shards = make_shards(world)
    ranges = reduce_ranges([oracle_key_ranges(s) for s in shards])
    parts = [[] for _ in range(world)]
    for r in range(world):
        states, counts = y.oracle_partial_mk(plan(), chunk_of(shards[r]),
                                             world, ranges)
        at = 0
        for p in range(world):
            seg = (type(states[0]) * max(counts[p], 1))()
            for i in range(counts[p]):
                seg[i] = states[at + i]
            parts[p].append((seg, counts[p]))
            at += counts[p]
    union = []
    for p in range(world):
        union += y.oracle_merge_mk(plan(), parts[p], ranges,
                                   col_types=[VT_INT64, VT_INT64, VT_INT64])
    big = y.Chunk(
        [y.encode_int64(np.concatenate([s[0] for s in shards]),
                        np.concatenate([s[1] for s in shards])),
         y.encode_int64(np.concatenate([s[2] for s in shards])),
         y.encode_int64(np.concatenate([s[3] for s in shards]),
                        np.concatenate([s[4] for s in shards]))],
        sum(len(s[0]) for s in shards))
    want, _ = y.oracle_execute(plan(), big)
    assert y.sort_rows(union) == y.sort_rows(want)
    # key-disjoint partitions
    keysets = [set((r2[0], r2[1]) for r2 in y.oracle_merge_mk(
        plan(), parts[p], ranges)) for p in range(world)]
    for a in range(world):
        for b in range(a + 1, world):
            assert not (keysets[a] & keysets[b])


@pytest.mark.gpu
def test_gpu_multikey_two_phase(cuda):
    shards = make_shards(1, n=150_000)
    ch = chunk_of(shards[0])
    dev = ch.c_device(cuda)
    ranges = y.gpu_key_ranges(plan(), dev)
    cap = 4 * 81 * 7 + 4096
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, st = y.gpu_partial_mk(plan(), dev, 1, states_t.data_ptr(), cap,
                                  ranges, max_groups_hint=4096)
    got, _ = y.gpu_merge_mk(plan(), states_t.data_ptr(), sum(counts), ranges,
                            col_types=[VT_INT64, VT_INT64, VT_INT64],
                            max_groups_hint=4096)
    want, _ = y.oracle_execute(plan(), chunk_of(shards[0]))
    assert y.sort_rows(got) == y.sort_rows(want)


@pytest.mark.gpu
def test_gpu_multikey_two_phase_matches_oracle_states(cuda):
    # cross-implementation: GPU partials merged by the ORACLE merge and
    # vice versa (packing must agree bit-for-bit)
    shards = make_shards(2, n=40_000)
    ranges = reduce_ranges([oracle_key_ranges(s) for s in shards])
    world = 2
    # rank 0 on GPU, rank 1 on oracle
    ch0 = chunk_of(shards[0])
    cap = 200_000
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts0, _ = y.gpu_partial_mk(plan(), ch0.c_device(cuda), world,
                                  states_t.data_ptr(), cap, ranges,
                                  max_groups_hint=65536)
    host_states = states_t.cpu().numpy().view(np.uint64)
    states1, counts1 = y.oracle_partial_mk(plan(), chunk_of(shards[1]),
                                           world, ranges)
    from ytsaurus_amd._abi import YtStateRow
    union = []
    at0 = 0
    at1 = 0
    for p in range(world):
        seg = (YtStateRow * max(counts0[p] + counts1[p], 1))()
        for i in range(counts0[p]):
            row = host_states[at0 + i]
            seg[i] = YtStateRow(key_bits=int(row[0]), meta=int(row[1]),
                                sum_bits=int(row[2]), row_count=int(row[3]))
        for i in range(counts1[p]):
            seg[counts0[p] + i] = states1[at1 + i]
        union += y.oracle_merge_mk(plan(), [(seg, counts0[p] + counts1[p])],
                                   ranges)
        at0 += counts0[p]
        at1 += counts1[p]
    big = y.Chunk(
        [y.encode_int64(np.concatenate([s[0] for s in shards]),
                        np.concatenate([s[1] for s in shards])),
         y.encode_int64(np.concatenate([s[2] for s in shards])),
         y.encode_int64(np.concatenate([s[3] for s in shards]),
                        np.concatenate([s[4] for s in shards]))],
        sum(len(s[0]) for s in shards))
    want, _ = y.oracle_execute(plan(), big)
    assert y.sort_rows(union) == y.sort_rows(want)


def test_oracle_multikey_avg_minmax_two_phase():
    """the mk exchange carries the same aggregate family as single-key:
    avg (exact counts) and min/max (raw-value states)"""
    from ytsaurus_amd._abi import AGG_MIN
    world = 2
    shards = make_shards(world, n=3000)
    for aggs in ([y.agg_avg(y.col(2)), y.agg_sum1()],
                 [y.agg_min(y.col(2)), y.agg_sum1()]):
        p = y.Plan(keys=[y.col(0), y.col(1)], aggs=list(aggs))
        ranges = reduce_ranges([oracle_key_ranges(s) for s in shards])
        parts = [[] for _ in range(world)]
        for r in range(world):
            states, counts = y.oracle_partial_mk(p, chunk_of(shards[r]),
                                                 world, ranges)
            at = 0
            for q in range(world):
                seg = (type(states[0]) * max(counts[q], 1))()
                for i in range(counts[q]):
                    seg[i] = states[at + i]
                parts[q].append((seg, counts[q]))
                at += counts[q]
        union = []
        for q in range(world):
            union += y.oracle_merge_mk(p, parts[q], ranges,
                                       col_types=[VT_INT64, VT_INT64,
                                                  VT_INT64])
        big = y.Chunk(
            [y.encode_int64(np.concatenate([s[0] for s in shards]),
                            np.concatenate([s[1] for s in shards])),
             y.encode_int64(np.concatenate([s[2] for s in shards])),
             y.encode_int64(np.concatenate([s[3] for s in shards]),
                            np.concatenate([s[4] for s in shards]))],
            sum(len(s[0]) for s in shards))
        want, _ = y.oracle_execute(p, big)
        wm = {(r[0], r[1]): r for r in want}
        assert len(union) == len(want)
        for row in union:
            w = wm[(row[0], row[1])]
            assert row[3] == w[3]
            if w[2] is None:
                assert row[2] is None
            elif isinstance(w[2], float):
                assert row[2] == pytest.approx(w[2], rel=1e-9)
            else:
                assert row[2] == w[2]


def test_coordinator_combine_totals_and_order():
    """the coordinator tail (DoCoordinateAndExecute, executor.cpp:761):
    key-disjoint front outputs concatenate, per-partition TOTALS rows
    re-fold into one, and the global ORDER BY ... LIMIT merges the
    per-partition-ordered unions — combined result == the single-pass
    query (GroupByCoordinatedWithTotals* family semantics)."""
    from ytsaurus_amd._abi import YtStateRow
    world = 3
    rng = np.random.default_rng(61)
    n = 9000
    k = rng.integers(-50, 50, n, dtype=np.int64)
    v = rng.integers(0, 10**6, n, dtype=np.int64)
    vn = (rng.random(n) < 0.1).astype(np.uint8)
    shards = []
    for r in range(world):
        sl = slice(r * n // world, (r + 1) * n // world)
        shards.append(y.Chunk([y.encode_int64(k[sl]),
                               y.encode_int64(v[sl], vn[sl])],
                              len(k[sl])))

    def planf():
        return y.Plan(keys=[y.col(0)],
                      aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                      order_by=[(1, True)], limit=12, with_totals=True)

    parts = [[] for _ in range(world)]
    for s in shards:
        states, counts = y.oracle_partial(
            y.Plan(keys=[y.col(0)],
                   aggs=[y.agg_sum(y.col(1)), y.agg_sum1()]), s, world)
        at = 0
        for p in range(world):
            seg = (YtStateRow * max(counts[p], 1))()
            for i in range(counts[p]):
                seg[i] = states[at + i]
            parts[p].append((seg, counts[p]))
            at += counts[p]
    results = [y.oracle_merge(planf(), parts[p]) for p in range(world)]
    combined = y.coordinate_results(planf(), results)
    big = y.Chunk([y.encode_int64(k), y.encode_int64(v, vn)], n)
    want, _ = y.oracle_execute(planf(), big)
    assert combined[-1] == want[-1]                  # totals row
    assert [r[1] for r in combined[:-1]] == [r[1] for r in want[:-1]]
    assert len(combined) == len(want)
