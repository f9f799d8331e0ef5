"""Two-item join chains (the reference's TMultiJoinParameters item list,
registry.cpp MultiJoinOpHelper:599-960; Multijoin ql_query_ut.cpp:5769 —
its full cross-product-on-both-items plain-scan shape stays refused: item 1
must have UNIQUE keys and dup item-0 plans must group). A later item may
key on an earlier item's appended column (snowflake chains)."""
import numpy as np
import pytest

import ytsaurus_amd as y


def enc(vals, nulls=None):
    return y.encode_int64(np.asarray(vals, dtype=np.int64),
                          None if nulls is None else
                          np.asarray(nulls, dtype=np.uint8))


def _model(fact_k, fact_v, j1, j2, j2_on_j1, is_left1, is_left2):
    """python reference: fact → dim1 (chains allowed) → dim2 (unique)"""
    d1 = {}
    for k, v in zip(*j1):
        d1.setdefault(k, []).append(v)
    d2 = dict(zip(*j2))
    out = {}
    for i, k in enumerate(fact_k):
        m1 = d1.get(k, [])
        if not m1:
            if not is_left1:
                continue
            m1 = [None]
        for b in m1:
            key2 = b if j2_on_j1 else k
            c = d2.get(key2)
            if c is None and key2 not in d2:
                if not is_left2:
                    continue
            g = out.setdefault(k, [0, 0, 0])
            g[0] += 1
            if b is not None:
                g[1] += b
            if c is not None:
                g[2] += c
    return sorted((k, g[0], g[1], g[2]) for k, g in out.items())


def build_case(seed, j2_on_j1, dup1=False, is_left1=False, is_left2=False,
               n=4000):
    rng = np.random.default_rng([88, seed])
    nk = 30
    fact_k = rng.integers(0, nk, n)
    fact_v = rng.integers(0, 100, n)
    # dim1: key -> b (dup1: some keys twice)
    d1k = list(range(nk - 4))
    if dup1:
        d1k += [0, 1, 2]
    d1b = [int(rng.integers(0, 15)) for _ in d1k]
    # dim2: keyed on b values (snowflake) or fact keys
    d2k = list(range(16)) if j2_on_j1 else list(range(nk - 2))
    d2c = [int(rng.integers(0, 1000)) for _ in d2k]

    fact = y.Chunk([enc(fact_k), enc(fact_v)], n)
    dim1 = y.Chunk([enc(d1k), enc(d1b)], len(d1k))
    dim2 = y.Chunk([enc(d2k), enc(d2c)], len(d2k))
    # plan columns: 0=fact_k 1=fact_v 2=dim1.b 3=dim2.c
    j1 = y.Join(dim1, primary_key_col=0, foreign_key_col=0, value_cols=[1],
                is_left=is_left1)
    j2 = y.Join(dim2, primary_key_col=(2 if j2_on_j1 else 0),
                foreign_key_col=0, value_cols=[1], is_left=is_left2)
    plan = y.Plan(keys=[y.col(0)],
                  aggs=[y.agg_sum1(), y.agg_sum(y.col(2)),
                        y.agg_sum(y.col(3))],
                  join=[j1, j2])
    want = _model(fact_k.tolist(), fact_v.tolist(), (d1k, d1b), (d2k, d2c),
                  j2_on_j1, is_left1, is_left2)
    return plan, fact, dim1, dim2, want


def norm(rows):
    return sorted((r[0], r[1], r[2] or 0, r[3] or 0) for r in rows)


@pytest.mark.parametrize("seed", range(4))
@pytest.mark.parametrize("j2_on_j1", [False, True])
def test_oracle_join_chain(seed, j2_on_j1):
    plan, fact, _, _, want = build_case(seed, j2_on_j1,
                                        dup1=bool(seed % 2),
                                        is_left1=seed >= 2,
                                        is_left2=bool((seed + 1) % 2))
    rows, _ = y.oracle_execute(plan, fact)
    assert norm(rows) == want, (seed, j2_on_j1)


def test_oracle_three_items_refused():
    plan, fact, dim1, dim2, _ = build_case(0, False)
    j3 = y.Join(dim2, primary_key_col=0, foreign_key_col=0, value_cols=[1])
    plan.join = plan.join + [j3]
    with pytest.raises(RuntimeError, match="more than two join items"):
        y.oracle_execute(plan, fact)


def test_oracle_dups_on_second_item_refused():
    rng = np.random.default_rng(5)
    fact = y.Chunk([enc([1, 2, 3]), enc([10, 20, 30])], 3)
    dupdim = y.Chunk([enc([1, 1, 2]), enc([5, 6, 7])], 3)
    unidim = y.Chunk([enc([1, 2, 3]), enc([8, 9, 10])], 3)
    j1 = y.Join(unidim, primary_key_col=0, foreign_key_col=0, value_cols=[1])
    j2 = y.Join(dupdim, primary_key_col=0, foreign_key_col=0, value_cols=[1])
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum1()], join=[j1, j2])
    with pytest.raises(RuntimeError, match="non-first join item"):
        y.oracle_execute(plan, fact)


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(4))
@pytest.mark.parametrize("j2_on_j1", [False, True])
def test_gpu_join_chain(cuda, seed, j2_on_j1):
    plan, fact, dim1, dim2, want = build_case(seed, j2_on_j1,
                                              dup1=bool(seed % 2),
                                              is_left1=seed >= 2,
                                              is_left2=bool((seed + 1) % 2))
    rows, _ = y.gpu_execute(plan, fact.c_device(cuda), max_groups_hint=512,
                            join_foreign=[dim1.c_device(cuda),
                                          dim2.c_device(cuda)])
    assert norm(rows) == want, (seed, j2_on_j1)


@pytest.mark.gpu
def test_gpu_join_chain_scan_project(cuda):
    # unique keys on both items: plain scan carries chained columns
    plan, fact, dim1, dim2, _ = build_case(0, True, dup1=False,
                                           is_left1=True, is_left2=True,
                                           n=500)
    plan2 = y.Plan(projects=[y.col(0), y.col(2), y.col(3)],
                   join=plan.join)
    got, _ = y.gpu_execute(plan2, fact.c_device(cuda),
                           join_foreign=[dim1.c_device(cuda),
                                         dim2.c_device(cuda)])
    want, _ = y.oracle_execute(plan2, fact)
    assert got == want
