"""Round-2 golden transcriptions, batch 3 (ql_query_ut.cpp): totals edge
semantics and the HAVING clause family. Expected rows transcribed verbatim
from the reference's own unit tests."""
import numpy as np
import pytest

import ytsaurus_amd as y


def enc(vals, nulls=None):
    return y.encode_int64(np.asarray(vals, dtype=np.int64),
                          None if nulls is None else
                          np.asarray(nulls, dtype=np.uint8))


def run(plan, chunk, cuda=None, hint=64):
    if cuda is None:
        rows, _ = y.oracle_execute(plan, chunk)
        return rows
    rows, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=hint)
    return rows


# GroupByWithTotalsNulls (:3989-4016): "x, sum(b) as t ... group by a % 2 as
# x with totals" over {a=1;b=10}, {b=20} → error "Null values are forbidden
# in group key" (a=null → x=null)
def totals_nulls_case():
    chunk = y.Chunk([enc([1, 0], [0, 1]), enc([10, 20])], 2)
    plan = y.Plan(keys=[y.col(0) % 2], aggs=[y.agg_sum(y.col(1))],
                  with_totals=True)
    return plan, chunk


def test_golden_totals_nulls_forbidden():
    plan, chunk = totals_nulls_case()
    with pytest.raises(RuntimeError, match="Null values are forbidden"):
        y.oracle_execute(plan, chunk)


# GroupByWithTotalsEmpty (:4018-4040): empty source with totals → EMPTY
# result (no totals row)
def test_golden_totals_empty():
    chunk = y.Chunk([enc([]), enc([])], 0)
    plan = y.Plan(keys=[y.col(0) % 2], aggs=[y.agg_sum(y.col(1))],
                  with_totals=True)
    rows, st = y.oracle_execute(plan, chunk)
    assert rows == []


# HavingClause1 (:4480): "a as x, sum(b) as t group by a having a = 1"
HAVING_SRC = ([1, 1, 2, 2], [10, 10, 20, 20])


def having_chunk():
    return y.Chunk([enc(HAVING_SRC[0]), enc(HAVING_SRC[1])], 4)


def test_golden_having1():
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1))],
                  having=y.col(0) == 1)
    assert run(plan, having_chunk()) == [(1, 20)]


# HavingClause2 (:4508): "... having sum(b) = 20"
def test_golden_having2():
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1))],
                  having=y.col(1) == 20)
    assert run(plan, having_chunk()) == [(1, 20)]


# HavingClause3 (:4536): "a as x group by a having sum(b) = 20" — the sum
# participates only in HAVING; our seam emits [key, aggs...] so the checked
# projection is the key column
def test_golden_having3():
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1))],
                  having=y.col(1) == 20)
    rows = run(plan, having_chunk())
    assert [r[0] for r in rows] == [1]


# GroupByWithLimitFirst (:4042-4070): "first(b) as f group by a limit 1"
# over rows a=1, b=0..9 → f=0 (the reference also asserts adaptive
# RowsRead=3, an ordered-reader early stop we do not replicate — we read
# whole resident chunks)
def test_golden_group_limit_first():
    chunk = y.Chunk([enc([1] * 10), enc(list(range(10)))], 10)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_first(y.col(1))])
    rows = run(plan, chunk)
    assert [(r[1],) for r in rows] == [(0,)]


@pytest.mark.gpu
def test_golden3_gpu(cuda):
    plan, chunk = totals_nulls_case()
    with pytest.raises(RuntimeError, match="Null values are forbidden"):
        y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=64)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1))],
                  having=y.col(0) == 1)
    assert run(plan, having_chunk(), cuda) == [(1, 20)]
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1))],
                  having=y.col(1) == 20)
    assert run(plan, having_chunk(), cuda) == [(1, 20)]
    chunk = y.Chunk([enc([1] * 10), enc(list(range(10)))], 10)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_first(y.col(1))])
    assert run(plan, chunk, cuda) == [(1, 0)]


# GroupByOrderBy3 (:2704-2758): filter + group + order by key + limit
def test_golden_group_order3():
    a = list(range(1, 10))
    b = [0, 1, 2] * 3
    chunk = y.Chunk([enc(a), enc(b)], 9)
    # "sum(a) as t, b where b = 0 group by b order by b limit 3" -> t=12;b=0
    plan = y.Plan(filter=y.col(1) == 0, keys=[y.col(1)],
                  aggs=[y.agg_sum(y.col(0))], order_by=[(0, False)], limit=3)
    rows, _ = y.oracle_execute(plan, chunk)
    assert rows == [(0, 12)]            # our emit order: [key, agg]
    # "... where b = 4 ..." -> empty
    plan = y.Plan(filter=y.col(1) == 4, keys=[y.col(1)],
                  aggs=[y.agg_sum(y.col(0))], order_by=[(0, False)], limit=3)
    rows, _ = y.oracle_execute(plan, chunk)
    assert rows == []


@pytest.mark.gpu
def test_golden_group_order3_gpu(cuda):
    a = list(range(1, 10))
    b = [0, 1, 2] * 3
    chunk = y.Chunk([enc(a), enc(b)], 9)
    plan = y.Plan(filter=y.col(1) == 0, keys=[y.col(1)],
                  aggs=[y.agg_sum(y.col(0))], order_by=[(0, False)], limit=3)
    rows, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=16)
    assert rows == [(0, 12)]
    plan = y.Plan(filter=y.col(1) == 4, keys=[y.col(1)],
                  aggs=[y.agg_sum(y.col(0))], order_by=[(0, False)], limit=3)
    rows, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=16)
    assert rows == []


# GroupByOrderBy2 first case (:2623-2668): string+int multi-key GROUP BY,
# ordered by the keys — oracle (string components refuse GPU composite
# packing loudly)
def test_golden_group_order2_string_multikey():
    a = list(range(1, 10))
    bs = ["a", "a", "b", "a", "b", "a", "b", "b", "a"]
    c = [1, 2, 3, 4, 1, 2, 3, 4, 1]
    chunk = y.Chunk([enc(a), y.encode_string(bs), enc(c)], 9)
    plan = y.Plan(keys=[y.col(1), y.col(2)], aggs=[y.agg_sum(y.col(0))],
                  order_by=[(0, False), (1, False)], limit=6)
    rows, _ = y.oracle_execute(plan, chunk)
    # reference rows (t,b,c) reordered to our [b, c, t] emit
    assert rows == [(b"a", 1, 10), (b"a", 2, 8), (b"a", 4, 4),
                    (b"b", 1, 5), (b"b", 3, 10), (b"b", 4, 8)]


# GroupByAlias (:3805-3838): "a % 3 as a, sum(a + b) as b group by a" — the
# alias SHADOWS the source column inside the aggregate (sum adds the
# aliased a%3, not the raw a: group 1 = (1+10)+(1+40)+(1+70) = 123). Our
# seam is post-parse, so the shadowing is restated explicitly.
def test_golden_group_by_alias():
    a = list(range(1, 10))
    b = [10 * i for i in a]
    chunk = y.Chunk([enc(a), enc(b)], 9)
    plan = y.Plan(keys=[y.col(0) % 3],
                  aggs=[y.agg_sum((y.col(0) % 3) + y.col(1))])
    rows = run(plan, chunk)
    assert y.sort_rows(rows) == y.sort_rows([(1, 123), (2, 156), (0, 180)])


@pytest.mark.gpu
def test_golden_group_by_alias_gpu(cuda):
    a = list(range(1, 10))
    b = [10 * i for i in a]
    chunk = y.Chunk([enc(a), enc(b)], 9)
    plan = y.Plan(keys=[y.col(0) % 3],
                  aggs=[y.agg_sum((y.col(0) % 3) + y.col(1))])
    rows = run(plan, chunk, cuda)
    assert y.sort_rows(rows) == y.sort_rows([(1, 123), (2, 156), (0, 180)])


# JoinEmpty (:5117-5164): odd b's join even b's -> nothing; group by a
# FOREIGN expression (c % 2)
def test_golden_join_empty(cuda=None):
    left = y.Chunk([enc([1, 3, 5, 7, 9]), enc([10, 30, 50, 70, 90])], 5)
    right = y.Chunk([enc([20, 40, 60, 80]), enc([2, 4, 6, 8])], 4)
    j = y.Join(right, primary_key_col=1, foreign_key_col=0, value_cols=[1])
    plan = y.Plan(keys=[y.col(2) % 2],
                  aggs=[y.agg_sum(y.col(0)), y.agg_sum(y.col(1))], join=j)
    rows, _ = y.oracle_execute(plan, left)
    assert rows == []


# JoinSimple2 (:5166-5203): unique keys, plain-scan join projection
def test_golden_join_simple2():
    left = y.Chunk([enc([1, 2])], 1 + 1)
    right = y.Chunk([enc([2, 1])], 2)
    j = y.Join(right, primary_key_col=0, foreign_key_col=0, value_cols=[])
    plan = y.Plan(projects=[y.col(0)], join=j)
    rows, _ = y.oracle_execute(plan, left)
    assert sorted(r[0] for r in rows) == [1, 2]


# JoinSimple3 (:5205-5242): DUPLICATE PRIMARY rows (a=1,1) x unique foreign
def test_golden_join_simple3():
    left = y.Chunk([enc([1, 1])], 2)
    right = y.Chunk([enc([2, 1])], 2)
    j = y.Join(right, primary_key_col=0, foreign_key_col=0, value_cols=[])
    plan = y.Plan(projects=[y.col(0)], join=j)
    rows, _ = y.oracle_execute(plan, left)
    assert [r[0] for r in rows] == [1, 1]


@pytest.mark.gpu
def test_golden_join_simple_gpu(cuda):
    left = y.Chunk([enc([1, 3, 5, 7, 9]), enc([10, 30, 50, 70, 90])], 5)
    right = y.Chunk([enc([20, 40, 60, 80]), enc([2, 4, 6, 8])], 4)
    j = y.Join(right, primary_key_col=1, foreign_key_col=0, value_cols=[1])
    plan = y.Plan(keys=[y.col(2) % 2],
                  aggs=[y.agg_sum(y.col(0)), y.agg_sum(y.col(1))], join=j)
    rows, _ = y.gpu_execute(plan, left.c_device(cuda), max_groups_hint=16,
                            join_foreign=right.c_device(cuda))
    assert rows == []
    left2 = y.Chunk([enc([1, 1])], 2)
    right2 = y.Chunk([enc([2, 1])], 2)
    j2 = y.Join(right2, primary_key_col=0, foreign_key_col=0, value_cols=[])
    plan2 = y.Plan(projects=[y.col(0)], join=j2)
    rows2, _ = y.gpu_execute(plan2, left2.c_device(cuda),
                             join_foreign=right2.c_device(cuda))
    assert [r[0] for r in rows2] == [1, 1]


# JoinNonPrefixColumns (:5659-5703) joins on a STRING key — a documented
# refusal this round (int64/uint64/boolean join keys only), asserted loud
def test_golden_join_string_key_refused():
    left = y.Chunk([y.encode_string(["a", "b", "c"]),
                    y.encode_string([None, None, None])], 3)
    right = y.Chunk([enc([1, 2, 3]), y.encode_string(["a", "b", "c"])], 3)
    j = y.Join(right, primary_key_col=0, foreign_key_col=1, value_cols=[0])
    plan = y.Plan(projects=[y.col(0), y.col(2)], join=j)
    with pytest.raises(RuntimeError, match="key columns"):
        y.oracle_execute(plan, left)
