"""avg() through the two-phase exchange: the 32-byte YtStateRow's
{sum_bits, meta bits8+ = EXACT non-null count} is precisely the
{count,sum} intermediate state the reference's coordinated avg carries
(GroupByWithAvgCoordinated ql_query_ut.cpp:2760-2794; avg finalize =
double(sum)/count)."""
import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd._abi import YtStateRow, VT_INT64, VT_DOUBLE


def enc(vals, nulls=None):
    return y.encode_int64(np.asarray(vals, dtype=np.int64),
                          None if nulls is None else
                          np.asarray(nulls, dtype=np.uint8))


def plan():
    return y.Plan(keys=[y.col(0)], aggs=[y.agg_avg(y.col(1)), y.agg_sum1()])


def test_golden_avg_coordinated():
    # GroupByWithAvgCoordinated: k=v=0..5 over 3 shards of 2 rows;
    # "avg(v) group by k" -> av = 0.0..5.0 (one row per group)
    shards = [y.Chunk([enc([2 * s, 2 * s + 1]), enc([2 * s, 2 * s + 1])], 2)
              for s in range(3)]
    world = 3
    parts = [[] for _ in range(world)]
    for s in shards:
        states, counts = y.oracle_partial(plan(), s, world)
        at = 0
        for p in range(world):
            seg = (YtStateRow * max(counts[p], 1))()
            for i in range(counts[p]):
                seg[i] = states[at + i]
            parts[p].append((seg, counts[p]))
            at += counts[p]
    union = []
    for p in range(world):
        union += y.oracle_merge(plan(), parts[p])
    assert sorted(r[1] for r in union) == [0.0, 1.0, 2.0, 3.0, 4.0, 5.0]
    assert all(isinstance(r[1], float) for r in union)


@pytest.mark.parametrize("double_arg", [False, True])
def test_oracle_avg_two_phase_fuzz(double_arg):
    world = 3
    shards = []
    for r in range(world):
        rng = np.random.default_rng(700 + r)
        n = 5000
        k = rng.integers(-60, 60, n, dtype=np.int64)
        kn = (rng.random(n) < 0.03).astype(np.uint8)
        vn = (rng.random(n) < 0.25).astype(np.uint8)
        if double_arg:
            v = rng.random(n) * 50 - 25
            vcol = y.encode_double(v, vn)
        else:
            v = rng.integers(-1000, 1000, n, dtype=np.int64)
            vcol = y.encode_int64(v, vn)
        shards.append((k, kn, v, vn,
                       y.Chunk([y.encode_int64(k, kn), vcol], n)))
    parts = [[] for _ in range(world)]
    for *_ignore, ch in shards:
        states, counts = y.oracle_partial(plan(), ch, world)
        at = 0
        for p in range(world):
            seg = (YtStateRow * max(counts[p], 1))()
            for i in range(counts[p]):
                seg[i] = states[at + i]
            parts[p].append((seg, counts[p]))
            at += counts[p]
    union = []
    for p in range(world):
        union += y.oracle_merge(plan(), parts[p])
    # single-pass over concatenated shards
    allk = np.concatenate([s[0] for s in shards])
    allkn = np.concatenate([s[1] for s in shards])
    allv = np.concatenate([s[2] for s in shards])
    allvn = np.concatenate([s[3] for s in shards])
    if double_arg:
        vcol = y.encode_double(allv, allvn)
    else:
        vcol = y.encode_int64(allv.astype(np.int64), allvn)
    big = y.Chunk([y.encode_int64(allk, allkn), vcol], len(allk))
    want, _ = y.oracle_execute(plan(), big)
    wm = {r[0]: r for r in want}
    assert len(union) == len(want)
    for k, av, cnt in union:
        _, wav, wc = wm[k]
        assert cnt == wc
        if wav is None:
            assert av is None
        else:
            assert av == pytest.approx(wav, rel=1e-9, abs=1e-12)


@pytest.mark.gpu
@pytest.mark.parametrize("double_arg", [False, True])
def test_gpu_avg_two_phase(cuda, double_arg):
    rng = np.random.default_rng(701)
    n = 150_000
    k = rng.integers(0, 997, n, dtype=np.int64)
    vn = (rng.random(n) < 0.3).astype(np.uint8)
    if double_arg:
        vcol = y.encode_double(rng.random(n) * 10, vn)
    else:
        vcol = y.encode_int64(rng.integers(0, 10**6, n, dtype=np.int64), vn)
    ch = y.Chunk([y.encode_int64(k), vcol], n)
    cap = 4 * 997 + 1024
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, st = y.gpu_partial(plan(), ch.c_device(cuda), 1,
                               states_t.data_ptr(), cap,
                               max_groups_hint=4096)
    got, _ = y.gpu_merge(plan(), states_t.data_ptr(), sum(counts),
                         max_groups_hint=4096,
                         col_types=[VT_INT64,
                                    VT_DOUBLE if double_arg else VT_INT64])
    want, _ = y.oracle_execute(plan(), ch)
    gm = {r[0]: r for r in got}
    assert len(got) == len(want)
    for kk, av, cnt in want:
        gk, gav, gc = gm[kk]
        assert gc == cnt
        if av is None:
            assert gav is None
        else:
            assert gav == pytest.approx(av, rel=1e-9, abs=1e-12)


@pytest.mark.gpu
def test_gpu_avg_states_merged_by_oracle(cuda):
    """cross-implementation: GPU avg partials merged by the ORACLE merge —
    the exact-count meta encoding must agree."""
    rng = np.random.default_rng(702)
    n = 30_000
    k = rng.integers(0, 51, n, dtype=np.int64)
    vn = (rng.random(n) < 0.2).astype(np.uint8)
    v = rng.integers(0, 1000, n, dtype=np.int64)
    ch = y.Chunk([y.encode_int64(k), y.encode_int64(v, vn)], n)
    cap = n + 16
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, _ = y.gpu_partial(plan(), ch.c_device(cuda), 1,
                              states_t.data_ptr(), cap, max_groups_hint=256)
    host = states_t.cpu().numpy().view(np.uint64)
    seg = (YtStateRow * max(counts[0], 1))()
    for i in range(counts[0]):
        row = host[i]
        seg[i] = YtStateRow(key_bits=int(row[0]), meta=int(row[1]),
                            sum_bits=int(row[2]), row_count=int(row[3]))
    union = y.oracle_merge(plan(), [(seg, counts[0])])
    want, _ = y.oracle_execute(plan(), ch)
    wm = {r[0]: r for r in want}
    assert len(union) == len(want)
    for kk, av, cnt in union:
        _, wav, wc = wm[kk]
        assert cnt == wc and av == pytest.approx(wav, rel=1e-12)


def mm_plan(is_max, col=1):
    f = y.agg_max if is_max else y.agg_min
    return y.Plan(keys=[y.col(0)], aggs=[f(y.col(col)), y.agg_sum1()])


@pytest.mark.parametrize("is_max", [False, True])
@pytest.mark.parametrize("double_arg", [False, True])
def test_oracle_minmax_two_phase(is_max, double_arg):
    """min/max through the exchange: states carry the RAW running value
    (meta bit1 types it); the merge re-min/maxes (udf/min.c merge =
    another min)."""
    world = 3
    shards = []
    for r in range(world):
        rng = np.random.default_rng(720 + r)
        n = 4000
        k = rng.integers(-40, 40, n, dtype=np.int64)
        kn = (rng.random(n) < 0.03).astype(np.uint8)
        vn = (rng.random(n) < 0.3).astype(np.uint8)
        if double_arg:
            v = rng.random(n) * 100 - 50
            vcol = y.encode_double(v, vn)
        else:
            v = rng.integers(-10**9, 10**9, n, dtype=np.int64)
            vcol = y.encode_int64(v, vn)
        shards.append((k, kn, v, vn,
                       y.Chunk([y.encode_int64(k, kn), vcol], n)))
    parts = [[] for _ in range(world)]
    for *_x, ch in shards:
        states, counts = y.oracle_partial(mm_plan(is_max), ch, world)
        at = 0
        for p in range(world):
            seg = (YtStateRow * max(counts[p], 1))()
            for i in range(counts[p]):
                seg[i] = states[at + i]
            parts[p].append((seg, counts[p]))
            at += counts[p]
    union = []
    for p in range(world):
        union += y.oracle_merge(mm_plan(is_max), parts[p])
    allk = np.concatenate([s[0] for s in shards])
    allkn = np.concatenate([s[1] for s in shards])
    allv = np.concatenate([s[2] for s in shards])
    allvn = np.concatenate([s[3] for s in shards])
    vcol = (y.encode_double(allv, allvn) if double_arg
            else y.encode_int64(allv.astype(np.int64), allvn))
    big = y.Chunk([y.encode_int64(allk, allkn), vcol], len(allk))
    want, _ = y.oracle_execute(mm_plan(is_max), big)
    assert y.sort_rows(union) == y.sort_rows(want)


@pytest.mark.gpu
@pytest.mark.parametrize("is_max", [False, True])
@pytest.mark.parametrize("double_arg", [False, True])
def test_gpu_minmax_two_phase(cuda, is_max, double_arg):
    rng = np.random.default_rng(721)
    n = 120_000
    k = rng.integers(0, 503, n, dtype=np.int64)
    vn = (rng.random(n) < 0.25).astype(np.uint8)
    if double_arg:
        vcol = y.encode_double(rng.random(n) * 20 - 10, vn)
    else:
        vcol = y.encode_int64(
            rng.integers(-10**12, 10**12, n, dtype=np.int64), vn)
    ch = y.Chunk([y.encode_int64(k), vcol], n)
    cap = 4 * 503 + 1024
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, _ = y.gpu_partial(mm_plan(is_max), ch.c_device(cuda), 1,
                              states_t.data_ptr(), cap, max_groups_hint=2048)
    got, _ = y.gpu_merge(mm_plan(is_max), states_t.data_ptr(), sum(counts),
                         max_groups_hint=2048,
                         col_types=[VT_INT64,
                                    VT_DOUBLE if double_arg else VT_INT64])
    want, _ = y.oracle_execute(mm_plan(is_max), ch)
    assert y.sort_rows(got) == y.sort_rows(want)


@pytest.mark.gpu
def test_gpu_minmax_states_cross_impl(cuda):
    """GPU min states merged by the ORACLE: the raw-value state encoding
    must agree across implementations."""
    rng = np.random.default_rng(722)
    n = 20_000
    k = rng.integers(0, 37, n, dtype=np.int64)
    v = rng.integers(-10**6, 10**6, n, dtype=np.int64)
    vn = (rng.random(n) < 0.2).astype(np.uint8)
    ch = y.Chunk([y.encode_int64(k), y.encode_int64(v, vn)], n)
    cap = n + 16
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, _ = y.gpu_partial(mm_plan(False), ch.c_device(cuda), 1,
                              states_t.data_ptr(), cap, max_groups_hint=128)
    host = states_t.cpu().numpy().view(np.uint64)
    seg = (YtStateRow * max(counts[0], 1))()
    for i in range(counts[0]):
        row = host[i]
        seg[i] = YtStateRow(key_bits=int(row[0]), meta=int(row[1]),
                            sum_bits=int(row[2]), row_count=int(row[3]))
    union = y.oracle_merge(mm_plan(False), [(seg, counts[0])])
    want, _ = y.oracle_execute(mm_plan(False), ch)
    assert y.sort_rows(union) == y.sort_rows(want)


def f_plan():
    return y.Plan(keys=[y.col(0)], aggs=[y.agg_first(y.col(1)), y.agg_sum1()])


def test_oracle_first_two_phase():
    """first() through the exchange: the first ARRIVING non-null state
    claims the slot (FirstIteration across tablets is likewise
    order-arbitrary; the test data has ONE distinct value per group so the
    pick is deterministic)."""
    world = 3
    shards = []
    for r in range(world):
        rng = np.random.default_rng(730 + r)
        n = 3000
        k = rng.integers(0, 40, n, dtype=np.int64)
        v = k * 7 + 1                      # value determined by the group
        vn = (rng.random(n) < 0.3).astype(np.uint8)
        shards.append(y.Chunk([enc(k), y.encode_int64(v, vn)], n))
    parts = [[] for _ in range(world)]
    for ch in shards:
        states, counts = y.oracle_partial(f_plan(), ch, world)
        at = 0
        for p in range(world):
            seg = (YtStateRow * max(counts[p], 1))()
            for i in range(counts[p]):
                seg[i] = states[at + i]
            parts[p].append((seg, counts[p]))
            at += counts[p]
    union = []
    for p in range(world):
        union += y.oracle_merge(f_plan(), parts[p])
    for k, fv, cnt in union:
        assert fv is None or fv == k * 7 + 1


@pytest.mark.gpu
def test_gpu_first_two_phase(cuda):
    rng = np.random.default_rng(731)
    n = 80_000
    k = rng.integers(0, 211, n, dtype=np.int64)
    v = k * 3 - 5
    vn = (rng.random(n) < 0.2).astype(np.uint8)
    ch = y.Chunk([y.encode_int64(k), y.encode_int64(v, vn)], n)
    cap = 4 * 211 + 1024
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, _ = y.gpu_partial(f_plan(), ch.c_device(cuda), 1,
                              states_t.data_ptr(), cap, max_groups_hint=1024)
    got, _ = y.gpu_merge(f_plan(), states_t.data_ptr(), sum(counts),
                         max_groups_hint=1024, col_types=[VT_INT64, VT_INT64])
    want, _ = y.oracle_execute(f_plan(), ch)
    assert y.sort_rows(got) == y.sort_rows(want)
