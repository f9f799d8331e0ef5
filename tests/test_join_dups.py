"""Duplicate-foreign-key joins: the reference's cross-product expansion
(registry.cpp MultiJoinOpHelper:599-960 sorted foreign sequences; golden
cardinality from ql_query_ut.cpp JoinSimple5 :5283-5329 — 3x3 same-key rows
join into 9, adapted to a GROUP BY since the plain-scan shape is gated this
round). VERDICT r1 missing item 2."""
import numpy as np
import pytest

import ytsaurus_amd as y


def enc(vals, nulls=None):
    return y.encode_int64(np.asarray(vals, dtype=np.int64),
                          None if nulls is None else
                          np.asarray(nulls, dtype=np.uint8))


def join_plan(fchunk, aggs, keys, is_left=False, **kw):
    j = y.Join(fchunk, primary_key_col=0, foreign_key_col=0,
               value_cols=[1], is_left=is_left)
    return y.Plan(keys=keys, aggs=aggs, join=j, **kw)


def run_both(plan, chunk, cuda=None):
    want, _ = y.oracle_execute(plan, chunk)
    if cuda is None:
        return want, None
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=512,
                           join_foreign=plan.join.chunk.c_device(cuda))
    return want, got


def test_join_simple5_cardinality():
    # JoinSimple5: 3 left x 3 right of one key -> 9 joined rows
    left = y.Chunk([enc([1, 1, 1]), enc([0, 0, 0])], 3)
    right = y.Chunk([enc([1, 1, 1]), enc([7, 8, 9])], 3)
    plan = join_plan(right, aggs=[y.agg_sum1(), y.agg_sum(y.col(2))],
                     keys=[y.col(0)])
    rows, _ = y.oracle_execute(plan, left)
    assert rows == [(1, 9, 3 * (7 + 8 + 9))]


def test_join_dup_mixed_keys():
    # k=1: 2 primary x 3 foreign = 6; k=2: 1x1; k=3: no match (INNER drop)
    left = y.Chunk([enc([1, 1, 2, 3]), enc([10, 20, 30, 40])], 4)
    right = y.Chunk([enc([1, 1, 1, 2]), enc([5, 6, 7, 8])], 4)
    plan = join_plan(right, aggs=[y.agg_sum1(), y.agg_sum(y.col(2)),
                                  y.agg_sum(y.col(1))],
                     keys=[y.col(0)])
    rows, _ = y.oracle_execute(plan, left)
    assert y.sort_rows(rows) == y.sort_rows([
        (1, 6, 2 * (5 + 6 + 7), 3 * (10 + 20)),
        (2, 1, 8, 30),
    ])


def test_join_dup_left_and_nulls():
    # LEFT: unmatched primary keeps one row with null foreign values;
    # null primary keys join ALL null-key foreign rows (eq-comparer
    # null == null, cg_fragment_compiler.cpp:425-447)
    left = y.Chunk([enc([1, 9, 0], [0, 0, 1]), enc([10, 20, 30])], 3)
    right = y.Chunk([enc([1, 0, 0], [0, 1, 1]), enc([5, 6, 7])], 3)
    plan = join_plan(right, aggs=[y.agg_sum1(), y.agg_sum(y.col(2))],
                     keys=[y.col(0)], is_left=True)
    rows, _ = y.oracle_execute(plan, left)
    assert y.sort_rows(rows) == y.sort_rows([
        (1, 1, 5),
        (9, 1, None),       # unmatched LEFT row
        (None, 2, 6 + 7),   # null key x 2 null-key foreign rows
    ])


def test_join_dup_order_by_refused():
    left = y.Chunk([enc([1, 1]), enc([1, 2])], 2)
    right = y.Chunk([enc([1, 1]), enc([3, 4])], 2)
    plan = join_plan(right, aggs=[y.agg_sum1()], keys=[y.col(0)],
                     order_by=[(0, False)], limit=10)
    with pytest.raises(RuntimeError, match="duplicate"):
        y.oracle_execute(plan, left)


def _rand_case(seed):
    rng = np.random.default_rng([77, seed])
    n = int(rng.integers(100, 4000))
    fn = int(rng.integers(10, 400))
    nkeys = int(rng.integers(2, 40))
    left_k = rng.integers(0, nkeys, n)
    left_kn = (rng.random(n) < 0.05).astype(np.uint8)
    left_v = rng.integers(-1000, 1000, n)
    right_k = rng.integers(0, nkeys, fn)
    right_kn = (rng.random(fn) < 0.05).astype(np.uint8)
    right_v = rng.integers(-1000, 1000, fn)
    right_vn = (rng.random(fn) < 0.1).astype(np.uint8)
    left = y.Chunk([enc(left_k, left_kn), enc(left_v)], n)
    right = y.Chunk([enc(right_k, right_kn), enc(right_v, right_vn)], fn)
    is_left = bool(seed % 2)
    plan = join_plan(right, aggs=[y.agg_sum1(), y.agg_sum(y.col(2)),
                                  y.agg_sum(y.col(1))],
                     keys=[y.col(0)], is_left=is_left)
    return plan, left


def _model_join(plan, left):
    """independent python cross-product group-by (checks the oracle too)"""
    import collections
    J = plan.join
    lk = [None if v is None else v for v in _col(left, 0)]
    lv = _col(left, 1)
    rk = _col(J.chunk_py, 0)
    rv = _col(J.chunk_py, 1)
    groups = collections.defaultdict(lambda: [0, 0, None, False, None, False])
    for i, k in enumerate(lk):
        matches = [j for j, fk in enumerate(rk) if fk == k]
        if not matches:
            if not plan.join.is_left:
                continue
            matches = [None]
        for m in matches:
            g = groups[k]
            g[0] += 1
            fv = None if m is None else rv[m]
            if fv is not None:
                g[2] = (g[2] or 0) + fv
                g[3] = True
            if lv[i] is not None:
                g[4] = (g[4] or 0) + lv[i]
                g[5] = True
    return [(k, g[0], g[2] if g[3] else None, g[4] if g[5] else None)
            for k, g in groups.items()]


def _col(chunk, i):
    # decode via oracle scan of identity projection
    plan = y.Plan(projects=[y.col(0), y.col(1)])
    rows, _ = y.oracle_execute(plan, chunk)
    return [r[i] for r in rows]


@pytest.mark.parametrize("seed", range(6))
def test_join_dup_fuzz_oracle_vs_model(seed):
    plan, left = _rand_case(seed)
    plan.join.chunk_py = plan.join.chunk
    want = _model_join(plan, left)
    rows, _ = y.oracle_execute(plan, left)
    def key(r):
        return (r[0] is None, r[0])
    got = sorted(((r[0], r[1], r[2], r[3]) for r in rows), key=key)
    want = sorted(want, key=key)
    assert got == want, seed


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(6))
def test_join_dup_fuzz_gpu_parity(cuda, seed):
    plan, left = _rand_case(seed)
    want, got = run_both(plan, left, cuda)
    assert y.sort_rows(got) == y.sort_rows(want), seed


@pytest.mark.gpu
def test_join_simple5_gpu(cuda):
    left = y.Chunk([enc([1, 1, 1]), enc([0, 0, 0])], 3)
    right = y.Chunk([enc([1, 1, 1]), enc([7, 8, 9])], 3)
    plan = join_plan(right, aggs=[y.agg_sum1(), y.agg_sum(y.col(2))],
                     keys=[y.col(0)])
    want, got = run_both(plan, left, cuda)
    assert got == [(1, 9, 3 * (7 + 8 + 9))]
    assert got == want
