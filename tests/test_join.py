"""Equi-join slice of SURVEY §8f row 4 (MultiJoinOpHelper,
cg_routines/registry.cpp:599-960): joined row = [primary columns...,
foreign columns...], INNER drops unmatched primaries, LEFT null-extends,
null keys join the null foreign key (the codegen eq-comparer treats
null == null, cg_fragment_compiler.cpp:425-447). This round: one join,
UNIQUE foreign keys (duplicates fail loudly); foreign values appear as
columns [P..) to filter/keys/aggs/order.
"""
import numpy as np
import pytest

import ytsaurus_amd as y


def test_join_oracle_smoke():
    # CPU-side sanity so the suite covers the join plumbing without a GPU
    rng = np.random.default_rng(80)
    pk, v, fkey, fval, _, chunk, fchunk, j = _mk(rng, n=5000, fn=100,
                                                 keyspace=150)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(2)), y.agg_sum1()],
                  join=j)
    rows, _ = y.oracle_execute(plan, chunk)
    fmap = dict(zip(fkey.tolist(), fval.tolist()))
    import collections
    want = collections.defaultdict(lambda: [0, 0])
    for k in pk.tolist():
        if k in fmap:
            want[k][0] += fmap[k]
            want[k][1] += 1
    assert len(rows) == len(want)
    for k, s_, c_ in rows:
        assert want[k] == [s_, c_]


def _mk(rng, n=50_000, fn=600, keyspace=800, left=False, vcols=1):
    pk = rng.integers(0, keyspace, n, dtype=np.int64)
    v = rng.integers(0, 10**6, n, dtype=np.int64)
    fkey = rng.permutation(np.arange(fn, dtype=np.int64))
    fval = rng.integers(-10**9, 10**9, fn, dtype=np.int64)
    fval2 = rng.integers(0, 100, fn, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(pk), y.encode_int64(v)], n)
    fchunk = y.Chunk([y.encode_int64(fkey), y.encode_int64(fval),
                      y.encode_int64(fval2)], fn)
    # value_cols index the FOREIGN chunk; they appear as plan cols [2..)
    j = y.Join(fchunk, 0, 0, [1, 2][:vcols], is_left=left)
    return pk, v, fkey, fval, fval2, chunk, fchunk, j


def _both(plan, chunk, fchunk, cuda, hint=8192):
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=hint,
                           join_foreign=fchunk.c_device(cuda))
    want, _ = y.oracle_execute(plan, chunk, nthreads=4)
    return got, want


@pytest.mark.gpu
def test_join_inner_group(cuda):
    rng = np.random.default_rng(81)
    pk, v, fkey, fval, _, chunk, fchunk, j = _mk(rng)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(2)), y.agg_sum1()],
                  join=j)
    got, want = _both(plan, chunk, fchunk, cuda)
    assert y.sort_rows(got) == y.sort_rows(want)
    assert all(r[0] < 600 for r in got)      # inner: unmatched keys dropped


@pytest.mark.gpu
def test_join_left_group(cuda):
    rng = np.random.default_rng(82)
    pk, v, fkey, fval, _, chunk, fchunk, j = _mk(rng, left=True)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(2)), y.agg_sum1()],
                  join=j)
    got, want = _both(plan, chunk, fchunk, cuda)
    assert y.sort_rows(got) == y.sort_rows(want)
    m = {r[0]: r for r in got}
    assert m[700][1] is None                  # unmatched: null foreign sum


@pytest.mark.gpu
def test_join_group_by_foreign_col(cuda):
    rng = np.random.default_rng(83)
    pk, v, fkey, fval, fval2, chunk, fchunk, j = _mk(rng, vcols=2)
    # group by the joined dimension attribute (col 3 = fval2)
    plan = y.Plan(keys=[y.col(3)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                  join=j)
    got, want = _both(plan, chunk, fchunk, cuda, hint=1024)
    assert y.sort_rows(got) == y.sort_rows(want)


@pytest.mark.gpu
def test_join_filter_on_foreign(cuda):
    rng = np.random.default_rng(84)
    pk, v, fkey, fval, _, chunk, fchunk, j = _mk(rng)
    plan = y.Plan(filter=y.col(2) > 0,
                  keys=[y.col(0)], aggs=[y.agg_sum1()], join=j)
    got, want = _both(plan, chunk, fchunk, cuda)
    assert y.sort_rows(got) == y.sort_rows(want)


@pytest.mark.gpu
def test_join_scan_project_and_topk(cuda):
    rng = np.random.default_rng(85)
    pk, v, fkey, fval, _, chunk, fchunk, j = _mk(rng, n=200_000)
    # scan+project with joined column
    plan = y.Plan(projects=[y.col(0), y.col(2)], join=j)
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda), out_capacity=250_000,
                           join_foreign=fchunk.c_device(cuda))
    want, _ = y.oracle_execute(plan, chunk)
    assert got == want
    # ORDER BY the joined column (k-selection path); ~333 primary rows share
    # each foreign value, so boundary ties are arbitrary: compare the ordered
    # key sequence and that every returned row exists in the joined data
    plan = y.Plan(projects=[y.col(0), y.col(2), y.col(1)],
                  order_by=[(1, True)], limit=50, join=j)
    got, want = _both(plan, chunk, fchunk, cuda)
    assert [r[1] for r in got] == [r[1] for r in want]
    fmap = dict(zip(fkey.tolist(), fval.tolist()))
    import collections
    src = collections.Counter(
        (int(pk[i]), int(fmap[int(pk[i])]), int(v[i]))
        for i in range(len(pk)) if int(pk[i]) in fmap)
    for r in got:
        assert src[tuple(r)] > 0
        src[tuple(r)] -= 1


@pytest.mark.gpu
def test_join_null_keys_match(cuda):
    # null primary keys join the null foreign key (eq-comparer null == null)
    pk = [1, None, 2, None, 3]
    fk = [None, 1, 2]
    fv = [111, 10, 20]
    n = len(pk)
    chunk = y.Chunk([y.encode_int64(np.array([x or 0 for x in pk], dtype=np.int64),
                                    np.array([x is None for x in pk], dtype=np.uint8)),
                     y.encode_int64(np.arange(n, dtype=np.int64))], n)
    fchunk = y.Chunk([y.encode_int64(np.array([x or 0 for x in fk], dtype=np.int64),
                                     np.array([x is None for x in fk], dtype=np.uint8)),
                      y.encode_int64(np.array(fv, dtype=np.int64))], len(fk))
    j = y.Join(fchunk, 0, 0, [1])
    plan = y.Plan(projects=[y.col(0), y.col(2)], join=j)
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda),
                           join_foreign=fchunk.c_device(cuda))
    want, _ = y.oracle_execute(plan, chunk)
    assert got == want
    assert (None, 111) in got                 # null joined null
    assert (3, None) not in got               # inner: key 3 dropped
    assert all(r[0] != 3 for r in got)


@pytest.mark.gpu
def test_join_duplicate_foreign_key_expands(cuda):
    # r2: duplicate foreign keys cross-product instead of erroring
    # (registry.cpp MultiJoinOpHelper; see test_join_dups.py for depth)
    rng = np.random.default_rng(86)
    pk, v, fkey, fval, _, chunk, fchunk, j = _mk(rng)
    dup = y.Chunk([y.encode_int64(np.array([5, 5, 7], dtype=np.int64)),
                   y.encode_int64(np.array([1, 2, 3], dtype=np.int64))], 3)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum1()],
                  join=y.Join(dup, 0, 0, [1]))
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda),
                           join_foreign=dup.c_device(cuda))
    want, _ = y.oracle_execute(plan, chunk)
    assert y.sort_rows(got) == y.sort_rows(want)


@pytest.mark.gpu
def test_join_with_order_having_totals(cuda):
    rng = np.random.default_rng(87)
    pk, v, fkey, fval, fval2, chunk, fchunk, j = _mk(rng, n=100_000, vcols=2)
    plan = y.Plan(keys=[y.col(3)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                  join=j, having=y.col(1) > 5 * 10**8,
                  order_by=[(1, True)], limit=10,
                  with_totals=True)
    got, want = _both(plan, chunk, fchunk, cuda, hint=1024)
    assert got == want
