"""WITH TOTALS — TotalsMode::BeforeHaving semantics (registry.cpp
TGroupByClosure::InsertTotals/FlushTotals:1556-1650): one extra output row
with null group keys whose aggregates cover ALL grouped rows, mirrored
from the reference's EStreamTag::Totals stream (appended last, flagged in
YtRowset.totals_row). Test semantics follow
TQueryEvaluateTest.GroupByCoordinatedWithTotalsNoLimitNoPrimaryKeyInGroupKey
(ql_query_ut.cpp:3432-3476): expected[gk] per group plus expected[null] =
grand total.
"""
import numpy as np
import pytest

import ytsaurus_amd as y


def _data(rng, n, nkeys=20):
    gk = rng.integers(0, nkeys, n, dtype=np.int64)
    v = rng.integers(0, 100, n, dtype=np.int64)
    return gk, v, y.Chunk([y.encode_int64(gk), y.encode_int64(v)], n)


def test_oracle_totals_matches_reference_shape():
    # ql_query_ut.cpp:3432: "gk, sum(v) from t group by gk with totals"
    rng = np.random.default_rng(51)
    gk, v, chunk = _data(rng, 5000)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1))], with_totals=True)
    rows, _ = y.oracle_execute(plan, chunk, nthreads=4)
    per = {int(k): int(v[gk == k].sum()) for k in np.unique(gk)}
    per[None] = int(v.sum())
    assert len(rows) == len(per)
    for k, s in rows:
        assert per.pop(k) == s
    assert not per


def test_oracle_totals_requires_group_by():
    rng = np.random.default_rng(52)
    _, _, chunk = _data(rng, 100)
    with pytest.raises(RuntimeError, match="GROUP BY"):
        y.oracle_execute(y.Plan(aggs=[y.agg_sum(y.col(1))], with_totals=True),
                         chunk)


def test_oracle_totals_with_null_key_group():
    # WITH TOTALS re-folds rows through the INTERMEDIATE stream, and the
    # reference forbids all-null group keys there
    # (registry.cpp ValidateGroupKeyIsNotNull:1460-1476; pinned by
    # GroupByWithTotalsNulls ql_query_ut.cpp:3989-4016)
    rng = np.random.default_rng(53)
    n = 2000
    gk = rng.integers(0, 5, n, dtype=np.int64)
    kn = (rng.random(n) < 0.2).astype(np.uint8)
    v = rng.integers(0, 100, n, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(gk, kn), y.encode_int64(v)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                  with_totals=True)
    with pytest.raises(RuntimeError, match="forbidden in group key"):
        y.oracle_execute(plan, chunk)


def test_oracle_totals_min_max_and_order():
    rng = np.random.default_rng(54)
    from ytsaurus_amd._abi import AGG_MIN, AGG_MAX
    n = 3000
    gk, v, chunk = _data(rng, n)
    plan = y.Plan(keys=[y.col(0)],
                  aggs=[(AGG_MIN, y.col(1)), (AGG_MAX, y.col(1)), y.agg_sum1()],
                  order_by=[(0, True)], limit=3, with_totals=True)
    rows, _ = y.oracle_execute(plan, chunk)
    # 3 ordered rows + totals; totals cover ALL groups, not the slice
    assert len(rows) == 4
    assert [r[0] for r in rows[:3]] == [19, 18, 17]
    assert rows[-1] == (None, int(v.min()), int(v.max()), n)


@pytest.mark.gpu
def test_totals_gpu_int(cuda):
    rng = np.random.default_rng(55)
    gk, v, chunk = _data(rng, 300_000, nkeys=1000)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                  with_totals=True)
    got, st = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=4096)
    want, _ = y.oracle_execute(plan, chunk, nthreads=4)
    assert got[-1] == want[-1] == (None, int(v.sum()), 300_000)
    assert y.sort_rows(got[:-1]) == y.sort_rows(want[:-1])


@pytest.mark.gpu
def test_totals_gpu_order_and_nullkeys(cuda):
    rng = np.random.default_rng(56)
    n = 200_000
    gk = rng.integers(0, 500, n, dtype=np.int64)
    v = rng.integers(0, 10**9, n, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(gk), y.encode_int64(v)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                  order_by=[(1, True)], limit=10, with_totals=True)
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=4096)
    want, _ = y.oracle_execute(plan, chunk)
    assert got == want
    assert got[-1] == (None, int(v.sum()), n)
    # a null group key under WITH TOTALS is forbidden on the GPU too
    kn = (rng.random(n) < 0.01).astype(np.uint8)
    chunk2 = y.Chunk([y.encode_int64(gk, kn), y.encode_int64(v)], n)
    with pytest.raises(RuntimeError, match="forbidden in group key"):
        y.gpu_execute(plan, chunk2.c_device(cuda), max_groups_hint=4096)


@pytest.mark.gpu
def test_totals_gpu_string_keys(cuda):
    rng = np.random.default_rng(57)
    n = 50_000
    keyset = ["s%03d" % i for i in range(200)]
    keys = [keyset[int(i)] for i in rng.integers(0, 200, n)]
    vals = rng.random(n)
    chunk = y.Chunk([y.encode_string(keys, max_segment_values=8192),
                     y.encode_double(vals, max_segment_values=8192)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                  with_totals=True)
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=1024)
    want, _ = y.oracle_execute(plan, chunk)
    assert got[-1][0] is None and got[-1][2] == n
    assert abs(got[-1][1] - vals.sum()) < 1e-6 * vals.sum()
    assert len(got) == len(want) == 201


@pytest.mark.gpu
def test_totals_gpu_two_phase(cuda):
    # bottom partials carry no totals; the front (merge) query computes them
    import ctypes as C
    rng = np.random.default_rng(58)
    gk, v, chunk = _data(rng, 100_000, nkeys=300)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])
    dev = chunk.c_device(cuda)
    cap = 4096
    states = cuda.zeros((2 * cap, 4), dtype=cuda.int64, device="cuda")
    counts, _ = y.gpu_partial(plan, dev, 2, states.data_ptr(), cap,
                              max_groups_hint=1024)
    mplan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                   with_totals=True)
    got, _ = y.gpu_merge(mplan, states.data_ptr(), sum(counts),
                         max_groups_hint=1024)
    want, _ = y.oracle_execute(mplan, chunk)
    assert y.sort_rows(got[:-1]) == y.sort_rows(want[:-1])
    assert got[-1] == want[-1] == (None, int(v.sum()), 100_000)


# --------- HAVING + totals-mode goldens (ql_query_ut.cpp:3916-3986) ---------
# "x, sum(b) as t FROM t where a > 1 group by a % 2 = 1 as x ..."
# Our seam emits [key, aggs] directly; having columns index that output row.

def _having_chunk():
    a = np.arange(1, 10, dtype=np.int64)
    b = a * 10
    return y.Chunk([y.encode_int64(a), y.encode_int64(b)], 9)


def _having_plan(having=None, with_totals=False, after=False):
    return y.Plan(filter=y.col(0) > 1,
                  keys=[(y.col(0) % 2) == 1],
                  aggs=[y.agg_sum(y.col(1))],
                  having=having, with_totals=with_totals,
                  totals_after_having=after)


def test_golden_with_totals():
    rows, _ = y.oracle_execute(_having_plan(with_totals=True), _having_chunk())
    assert rows == [(False, 200), (True, 240), (None, 440)]


def test_golden_having_then_totals():
    # "having t > 200 with totals" → AfterHaving: totals over survivors
    plan = _having_plan(having=y.col(1) > 200, with_totals=True, after=True)
    rows, _ = y.oracle_execute(plan, _having_chunk())
    assert rows == [(True, 240), (None, 240)]


def test_golden_totals_then_having():
    # "with totals having t > 200" → BeforeHaving: totals over all groups
    plan = _having_plan(having=y.col(1) > 200, with_totals=True)
    rows, _ = y.oracle_execute(plan, _having_chunk())
    assert rows == [(True, 240), (None, 440)]
    plan = _having_plan(having=y.col(1) < 220, with_totals=True)
    rows, _ = y.oracle_execute(plan, _having_chunk())
    assert rows == [(False, 200), (None, 440)]


def test_having_requires_group_by():
    with pytest.raises(RuntimeError, match="GROUP BY"):
        y.oracle_execute(y.Plan(projects=[y.col(0)], having=y.col(0) > 1),
                         _having_chunk())


@pytest.mark.gpu
def test_golden_having_totals_gpu(cuda):
    chunk = _having_chunk()
    for kwargs, want in [
        (dict(with_totals=True),
         [(False, 200), (True, 240), (None, 440)]),
        (dict(having=y.col(1) > 200, with_totals=True, after=True),
         [(True, 240), (None, 240)]),
        (dict(having=y.col(1) > 200, with_totals=True),
         [(True, 240), (None, 440)]),
        (dict(having=y.col(1) < 220, with_totals=True),
         [(False, 200), (None, 440)]),
    ]:
        got, _ = y.gpu_execute(_having_plan(**kwargs), chunk.c_device(cuda),
                               max_groups_hint=64)
        assert got[-1] == want[-1]                       # totals row
        assert y.sort_rows(got[:-1]) == y.sort_rows(want[:-1])


@pytest.mark.gpu
def test_having_gpu_large(cuda):
    rng = np.random.default_rng(60)
    n = 400_000
    gk = rng.integers(0, 2000, n, dtype=np.int64)
    v = rng.integers(0, 10**6, n, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(gk), y.encode_int64(v)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                  having=y.col(1) > 10**8, order_by=[(1, True)], limit=20,
                  with_totals=True, totals_after_having=True)
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=4096)
    want, _ = y.oracle_execute(plan, chunk, nthreads=4)
    assert got == want
