"""first() and avg aggregates (SURVEY §2 aggregate
registry: udf/first.c + registry.cpp FirstIteration:3642-3663,
builtin_function_profiler.cpp avg codegen :1483-1607).

Goldens transcribed from the reference's own tests:
ql_query_ut.cpp AverageAgg (:8616-8639), AverageAgg2 (:8668-8703),
AverageAgg3 (:8705-8732).
"""
import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd._abi import AGG_MIN, AGG_MAX, AGG_FIRST, AGG_AVG


def chunk_i64(*cols):
    n = len(cols[0][0])
    enc = []
    for vals, nulls in cols:
        enc.append(y.encode_int64(np.asarray(vals, dtype=np.int64),
                                  None if nulls is None else
                                  np.asarray(nulls, dtype=np.uint8)))
    return y.Chunk(enc, n)


# ---- goldens (oracle = the restated reference algorithm) ----

def test_average_agg_golden():
    # ql_query_ut.cpp:8616-8639 AverageAgg: avg(a) group by 1 -> x=24.2
    ch = chunk_i64(([3, 53, 8, 24, 33], None))
    plan = y.Plan(keys=[y.lit(1)], aggs=[y.agg_avg(y.col(0))],
                  projects=[y.col(1)])
    rows, _ = y.oracle_execute(plan, ch)
    assert rows == [(24.2,)]


def test_average_agg2_golden():
    # ql_query_ut.cpp:8668-8703 AverageAgg2
    a = [3, 53, 8, 24, 33, 33, 23, 33]
    b = [3, 2, 5, 7, 4, 3, 0, 8]
    c = [1, 3, 32, 4, 9, 43, 0, 2]
    ch = chunk_i64((a, None), (b, None), (c, None))
    plan = y.Plan(keys=[y.col(1) % 2],
                  aggs=[y.agg_avg(y.col(0)), (AGG_MAX, y.col(2)),
                        y.agg_avg(y.col(2)), (AGG_MIN, y.col(0))],
                  projects=[y.col(1), y.col(0), y.col(2), y.col(3), y.col(4)])
    rows, _ = y.oracle_execute(plan, ch)
    got = sorted(rows, key=lambda r: r[1])
    assert got == [(35.5, 0, 9, 3.5, 23), (17.0, 1, 43, 20.0, 3)]


def test_average_agg3_golden():
    # ql_query_ut.cpp:8705-8732 AverageAgg3: avg over doubles with nulls
    av = np.array([3.0, 0.0, 0.0, 7.0])
    an = np.array([0, 1, 1, 0], dtype=np.uint8)
    bv = np.array([1, 1, 0, 1], dtype=np.int64)
    ch = y.Chunk([y.encode_double(av, an), y.encode_int64(bv)], 4)
    plan = y.Plan(keys=[y.col(1)], aggs=[y.agg_avg(y.col(0))])
    rows, _ = y.oracle_execute(plan, ch)
    assert sorted(rows) == [(0, None), (1, 5.0)]


def test_first_oracle_row_order():
    # FirstIteration keeps the first NON-NULL in scan order
    v = np.array([0, 10, 20, 30, 40, 50], dtype=np.int64)
    vn = np.array([1, 0, 0, 0, 0, 0], dtype=np.uint8)   # first row null
    k = np.array([7, 7, 7, 8, 8, 8], dtype=np.int64)
    ch = y.Chunk([y.encode_int64(k), y.encode_int64(v, vn)], 6)
    plan = y.Plan(keys=[y.col(0)], aggs=[(AGG_FIRST, y.col(1))])
    rows, _ = y.oracle_execute(plan, ch)
    assert sorted(rows) == [(7, 10), (8, 30)]


def test_avg_with_totals_refused():
    ch = chunk_i64(([1, 2], None))
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_avg(y.col(0))],
                  with_totals=True)
    with pytest.raises(RuntimeError):
        y.oracle_execute(plan, ch)


# ---- GPU parity ----

@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(3))
def test_avg_gpu_parity_int(cuda, seed):
    rng = np.random.default_rng([41, seed])
    n = 120_000
    k = rng.integers(0, 500, n).astype(np.int64)
    v = rng.integers(-10**12, 10**12, n).astype(np.int64)
    vn = (rng.random(n) < 0.1).astype(np.uint8)
    ch = y.Chunk([y.encode_int64(k), y.encode_int64(v, vn)], n)
    plan = y.Plan(keys=[y.col(0)],
                  aggs=[y.agg_avg(y.col(1)), y.agg_sum1()])
    got, _ = y.gpu_execute(plan, ch.c_device(cuda), max_groups_hint=1024)
    want, _ = y.oracle_execute(plan, ch)
    gm = {r[0]: r for r in got}
    assert len(got) == len(want)
    for key, avgv, cnt in want:
        gk, ga, gc = gm[key]
        assert gc == cnt
        if avgv is None:
            assert ga is None
        else:
            assert ga == pytest.approx(avgv, rel=1e-12)


@pytest.mark.gpu
def test_avg_gpu_parity_double(cuda):
    rng = np.random.default_rng(42)
    n = 80_000
    k = rng.integers(0, 200, n).astype(np.int64)
    v = rng.random(n) * 1000
    vn = (rng.random(n) < 0.05).astype(np.uint8)
    ch = y.Chunk([y.encode_int64(k), y.encode_double(v, vn)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_avg(y.col(1))])
    got, _ = y.gpu_execute(plan, ch.c_device(cuda), max_groups_hint=1024)
    want, _ = y.oracle_execute(plan, ch)
    gm = {r[0]: r[1] for r in got}
    assert len(got) == len(want)
    for key, avgv in want:
        if avgv is None:
            assert gm[key] is None
        else:
            # double adds commute only approximately across thread order
            assert gm[key] == pytest.approx(avgv, rel=1e-9)


@pytest.mark.gpu
def test_first_gpu_parity_deterministic(cuda):
    # one distinct value per group (plus nulls): the pick is forced, so
    # GPU (arbitrary scan order) must equal the oracle (row order)
    rng = np.random.default_rng(43)
    n = 100_000
    k = rng.integers(0, 300, n).astype(np.int64)
    v = k * 7 - 3
    vn = (rng.random(n) < 0.3).astype(np.uint8)
    ch = y.Chunk([y.encode_int64(k), y.encode_int64(v, vn)], n)
    plan = y.Plan(keys=[y.col(0)],
                  aggs=[(AGG_FIRST, y.col(1)), y.agg_sum1()])
    got, _ = y.gpu_execute(plan, ch.c_device(cuda), max_groups_hint=1024)
    want, _ = y.oracle_execute(plan, ch)
    # all-null groups may differ in which value... no: single distinct value
    assert y.sort_rows(got) == y.sort_rows(want)


@pytest.mark.gpu
def test_first_gpu_membership(cuda):
    # several distinct values per group: GPU's pick must be a member of
    # the group's non-null value set (registry.cpp FirstIteration picks
    # scan-order-first; order across devices is unspecified)
    rng = np.random.default_rng(44)
    n = 50_000
    k = rng.integers(0, 100, n).astype(np.int64)
    v = rng.integers(0, 10**9, n).astype(np.int64)
    ch = y.Chunk([y.encode_int64(k), y.encode_int64(v)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[(AGG_FIRST, y.col(1))])
    got, _ = y.gpu_execute(plan, ch.c_device(cuda), max_groups_hint=1024)
    sets = {}
    for kk, vv in zip(k.tolist(), v.tolist()):
        sets.setdefault(kk, set()).add(vv)
    assert len(got) == len(sets)
    for kk, vv in got:
        assert vv in sets[kk]


@pytest.mark.gpu
def test_first_string_refused_on_gpu(cuda):
    keys = ["a", "b", "a"]
    ch = y.Chunk([y.encode_int64(np.array([1, 2, 1], dtype=np.int64)),
                  y.encode_string(keys)], 3)
    plan = y.Plan(keys=[y.col(0)], aggs=[(AGG_FIRST, y.col(1))])
    with pytest.raises(RuntimeError, match="oracle-only|unsupported"):
        y.gpu_execute(plan, ch.c_device(cuda), max_groups_hint=64)
