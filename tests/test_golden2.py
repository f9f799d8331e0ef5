"""Round-2 golden transcriptions from the reference's own evaluator tests
(VERDICT r1 item 7). Each case cites its ql_query_ut.cpp source. The
coordinated tests' implicit key order (sorted dynamic tables) is expressed
here as an explicit ORDER BY over the group keys — same rowset contract
(EvaluateCoordinatedGroupBy feeds a merged reader into the same front
query; test_evaluate.cpp:753-860).
"""
import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd._abi import AGG_MIN


def chunk(cols, n):
    return y.Chunk(cols, n)


def enc(vals, nulls=None):
    return y.encode_int64(np.asarray(vals, dtype=np.int64),
                          None if nulls is None else
                          np.asarray(nulls, dtype=np.uint8))


def cases():
    out = []

    # ComplexWithAliases (ql_query_ut.cpp:4127-4160):
    # a % 2 as x, sum(b) + x as t WHERE a > 1 GROUP BY x
    a = list(range(1, 10))
    b = [10 * i for i in a]
    out.append(dict(
        name="ComplexWithAliases",
        cols=[enc(a), enc(b)], n=9,
        plan=dict(filter=y.col(0) > 1,
                  keys=[y.col(0) % 2],
                  aggs=[y.agg_sum(y.col(1))],
                  projects=[y.col(0), y.col(1) + y.col(0)]),
        expected=[(0, 200), (1, 241)],
        ordered=False))

    # GroupByCoordinatedWithAggregates1 (:3261-3296):
    # k1, sum(v) group by k1 offset 1 limit 5 (key order = table sort order)
    k1 = [0, 1, 2, 3, 3, 4]
    k2 = [0, 0, 0, 0, 1, 0]
    v = [0, 1, 2, 3, 4, 5]
    out.append(dict(
        name="CoordinatedWithAggregates1",
        cols=[enc(k1), enc(k2), enc(v)], n=6,
        plan=dict(keys=[y.col(0)], aggs=[y.agg_sum(y.col(2))],
                  order_by=[(0, False)], limit=5, offset=1),
        expected=[(1, 1), (2, 2), (3, 7), (4, 5)],
        ordered=True))

    # GroupByCoordinatedWithAggregates2 (:3298-3334):
    # k0, v2, min(v3) group by k0, v2 limit 1
    out.append(dict(
        name="CoordinatedWithAggregates2",
        cols=[enc([1, 1, 1, 1]), enc([1, 2, 3, 4]),
              enc([1, 2, 2, 1]), enc([42, 1, 1, 0])], n=4,
        plan=dict(keys=[y.col(0), y.col(2)],
                  aggs=[(AGG_MIN, y.col(3))],
                  order_by=[(0, False), (1, False)], limit=1),
        expected=[(1, 1, 0)],
        ordered=True))

    # GroupByCoordinatedWithAggregates3 (:3336-3383):
    # k1 = i/10, v1 = i over i in [0,100); three limit/offset variants
    i = np.arange(100)
    c3 = [enc(i // 10), enc(i % 10), enc(i)]
    for name, lim, off, want in [
            ("CoordinatedWithAggregates3a", 1, 0, [(0, 45)]),
            ("CoordinatedWithAggregates3b", 2, 0, [(0, 45), (1, 145)]),
            ("CoordinatedWithAggregates3c", 1, 1, [(1, 145)])]:
        out.append(dict(
            name=name, cols=c3, n=100,
            plan=dict(keys=[y.col(0)], aggs=[y.agg_sum(y.col(2))],
                      order_by=[(0, False)], limit=lim, offset=off),
            expected=want, ordered=True))

    # GroupByCoordinatedWithAggregates4 (:3385-3431): nullable group key
    # k0, min(k2) group by k0 — k0 null for 9 rows, 0 for 2
    k0 = [0] * 9 + [0, 0]
    k0n = [1] * 9 + [0, 0]
    k2 = [9, 8, 7, 6, 5, 4, 3, 2, 1, 0, 5]
    out.append(dict(
        name="CoordinatedWithAggregates4",
        cols=[enc(k0, k0n), enc(list(range(1, 10)) + [0, 1]), enc(k2)], n=11,
        plan=dict(keys=[y.col(0)], aggs=[(AGG_MIN, y.col(2))]),
        expected=[(None, 1), (0, 0)],
        ordered=False))

    # AvgCoordinated (:2760-2790): avg(v) group by k, one value per key
    out.append(dict(
        name="AvgCoordinated",
        cols=[enc(range(6)), enc(range(6))], n=6,
        plan=dict(keys=[y.col(0)], aggs=[y.agg_avg(y.col(1))],
                  projects=[y.col(1)]),
        expected=[(0.0,), (1.0,), (2.0,), (3.0,), (4.0,), (5.0,)],
        ordered=False))

    return out


CASES = cases()


def run_plan(spec):
    return y.Plan(**spec)


@pytest.mark.parametrize("case", CASES, ids=[c["name"] for c in CASES])
def test_golden2_oracle(case):
    rows, _ = y.oracle_execute(run_plan(case["plan"]),
                               chunk(case["cols"], case["n"]))
    if case["ordered"]:
        assert rows == case["expected"]
    else:
        assert y.sort_rows(rows) == y.sort_rows(case["expected"])


@pytest.mark.gpu
@pytest.mark.parametrize("case", CASES, ids=[c["name"] for c in CASES])
def test_golden2_gpu(case, cuda):
    got, _ = y.gpu_execute(run_plan(case["plan"]),
                           chunk(case["cols"], case["n"]).c_device(cuda),
                           max_groups_hint=256)
    if case["ordered"]:
        assert got == case["expected"]
    else:
        assert y.sort_rows(got) == y.sort_rows(case["expected"])
