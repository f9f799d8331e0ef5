"""Join in the BOTTOM query of the coordinated split: the reference keeps
the JoinClause in the bottom query (engine/coordinator.cpp:130-170 — the
dimension table is resolvable on every node), so each rank builds the
foreign hash table from its own copy of the dimension chunk, probes it
during the partial scan (incl. duplicate-key cross-product expansion), and
the exchanged YtStateRow format is unchanged."""
import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd._abi import YtStateRow, VT_INT64


def enc(vals, nulls=None):
    return y.encode_int64(np.asarray(vals, dtype=np.int64),
                          None if nulls is None else
                          np.asarray(nulls, dtype=np.uint8))


def make_dim(nkeys=37, seed=5):
    rng = np.random.default_rng(seed)
    # duplicate foreign keys included (two rows for some keys)
    keys = np.concatenate([np.arange(nkeys), rng.integers(0, nkeys, 6)])
    vals = rng.integers(-100, 100, len(keys))
    return y.Chunk([enc(keys), enc(vals)], len(keys))


def make_shards(world, n=5000, nkeys=37):
    shards = []
    for r in range(world):
        rng = np.random.default_rng(800 + r)
        k = rng.integers(-2, nkeys + 2, n)          # some unmatched
        kn = (rng.random(n) < 0.04).astype(np.uint8)
        v = rng.integers(-1000, 1000, n)
        shards.append(y.Chunk([enc(k, kn), enc(v)], n))
    return shards


def plan_of(dim, is_left=False):
    j = y.Join(dim, primary_key_col=0, foreign_key_col=0, value_cols=[1],
               is_left=is_left)
    # the 32-byte YtStateRow carries ONE sum state + the row count
    return y.Plan(keys=[y.col(0)],
                  aggs=[y.agg_sum(y.col(2)), y.agg_sum1()],
                  join=j)


@pytest.mark.parametrize("is_left", [False, True])
def test_oracle_two_phase_join(is_left):
    world = 3
    dim = make_dim()
    shards = make_shards(world)
    parts = [[] for _ in range(world)]
    for r in range(world):
        states, counts = y.oracle_partial(plan_of(dim, is_left), shards[r],
                                          world)
        at = 0
        for p in range(world):
            seg = (YtStateRow * max(counts[p], 1))()
            for i in range(counts[p]):
                seg[i] = states[at + i]
            parts[p].append((seg, counts[p]))
            at += counts[p]
    union = []
    for p in range(world):
        union += y.oracle_merge(plan_of(dim, is_left), parts[p])
    # single-pass over the concatenated shards
    allk = np.concatenate([
        np.array([(-10**9 if v is None else v)
                  for v in _col(s, 0)], dtype=np.int64) for s in shards])
    allkn = np.concatenate([
        np.array([v is None for v in _col(s, 0)], dtype=np.uint8)
        for s in shards])
    allv = np.concatenate([
        np.array(_col(s, 1), dtype=np.int64) for s in shards])
    big = y.Chunk([enc(allk, allkn), enc(allv)], len(allk))
    want, _ = y.oracle_execute(plan_of(dim, is_left), big)
    assert y.sort_rows(union) == y.sort_rows(want)


def _col(chunk, i):
    plan = y.Plan(projects=[y.col(0), y.col(1)])
    rows, _ = y.oracle_execute(plan, chunk)
    return [r[i] for r in rows]


def test_oracle_partial_mk_join_refused():
    dim = make_dim()
    j = y.Join(dim, primary_key_col=0, foreign_key_col=0, value_cols=[1])
    plan = y.Plan(keys=[y.col(0), y.col(1)], aggs=[y.agg_sum1()], join=j)
    with pytest.raises(RuntimeError, match="multi-key"):
        y.oracle_partial_mk(plan, make_shards(1)[0], 2,
                            ([0, 0], [10, 10]))


@pytest.mark.gpu
@pytest.mark.parametrize("is_left", [False, True])
def test_gpu_two_phase_join(cuda, is_left):
    dim = make_dim()
    shards = make_shards(1, n=60_000)
    ch = shards[0]
    cap = 4 * 45 + 1024
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, st = y.gpu_partial(plan_of(dim, is_left), ch.c_device(cuda), 1,
                               states_t.data_ptr(), cap,
                               max_groups_hint=4096,
                               join_foreign=dim.c_device(cuda))
    got, _ = y.gpu_merge(plan_of(dim, is_left), states_t.data_ptr(),
                         sum(counts), max_groups_hint=4096,
                         col_types=[VT_INT64])
    want, _ = y.oracle_execute(plan_of(dim, is_left), ch)
    assert y.sort_rows(got) == y.sort_rows(want)
