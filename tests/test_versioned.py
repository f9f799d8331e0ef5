"""Versioned scan-format slice: the synthetic encoder
must round-trip through the restated reader with exact
rowset_builder.cpp:1042-1166 visibility semantics (produceAll=false):
per row, timestamps DESC; deleteTs = latest delete <= T; a row is visible
iff it has a write <= T newer than deleteTs; the column value is the
newest such write's value. The model here is an independent pure-Python
restatement of the same rule.
"""
import numpy as np
import pytest

import ytsaurus_amd as y


def _gen(rng, n, max_writes=4, max_deletes=2, null_frac=0.1, ts_space=1000):
    wpr = rng.integers(0, max_writes + 1, n).astype(np.uint32)
    dpr = rng.integers(0, max_deletes + 1, n).astype(np.uint32)
    wts, dts, vals, vnul = [], [], [], []
    rows = []
    for r in range(n):
        w = np.sort(rng.choice(ts_space, size=int(wpr[r]), replace=False))[::-1] + 1
        d = np.sort(rng.choice(ts_space, size=int(dpr[r]), replace=False))[::-1] + 1
        v = rng.integers(-10**9, 10**9, int(wpr[r]))
        nl = (rng.random(int(wpr[r])) < null_frac).astype(np.uint8)
        wts.extend(w.tolist())
        dts.extend(d.tolist())
        vals.extend(v.tolist())
        vnul.extend(nl.tolist())
        rows.append((w.tolist(), d.tolist(), v.tolist(), nl.tolist()))
    col = y.encode_versioned_int64(wpr, np.array(wts, dtype=np.uint64),
                                   np.array(vals, dtype=np.int64),
                                   np.array(vnul, dtype=np.uint8),
                                   dpr, np.array(dts, dtype=np.uint64))
    return rows, col


def _model(rows, T):
    vals, vis = [], []
    for w, d, v, nl in rows:
        delete_ts = next((ts for ts in d if ts <= T), 0)
        visible = [(i, ts) for i, ts in enumerate(w) if delete_ts < ts <= T]
        if not visible:
            vals.append(None)
            vis.append(False)
            continue
        i, _ = visible[0]          # DESC order: first qualifying is newest
        vals.append(None if nl[i] else int(v[i]))
        vis.append(True)
    return vals, vis


@pytest.mark.parametrize("seed", range(12))
def test_versioned_read_matches_model(seed):
    rng = np.random.default_rng([7, seed])
    n = int(rng.choice([1, 50, 3000]))
    rows, col = _gen(rng, n)
    for T in [0, 1, 100, 500, 999, 1000, 10**15]:
        got_vals, got_vis = y.oracle_versioned_read(col, T)
        want_vals, want_vis = _model(rows, T)
        assert got_vis == want_vis, f"T={T}"
        assert got_vals == want_vals, f"T={T}"


def test_versioned_multi_segment():
    rng = np.random.default_rng(8)
    rows, col = _gen(rng, 1000)
    rows2, col2 = rows, None
    # re-encode with tiny segments: identical reads
    wpr = np.array([len(r[0]) for r in rows], dtype=np.uint32)
    dpr = np.array([len(r[1]) for r in rows], dtype=np.uint32)
    wts = np.array([t for r in rows for t in r[0]], dtype=np.uint64)
    dts = np.array([t for r in rows for t in r[1]], dtype=np.uint64)
    vals = np.array([x for r in rows for x in r[2]], dtype=np.int64)
    vnul = np.array([x for r in rows for x in r[3]], dtype=np.uint8)
    col2 = y.encode_versioned_int64(wpr, wts, vals, vnul, dpr, dts,
                                    max_rows_per_segment=64)
    assert col2._c.ts_seg_count == (1000 + 63) // 64
    for T in [0, 250, 750, 10**9]:
        assert y.oracle_versioned_read(col, T) == \
               y.oracle_versioned_read(col2, T)


def test_versioned_delete_shadowing():
    # one row: writes at 30, 20, 10; delete at 25
    col = y.encode_versioned_int64(
        np.array([3], dtype=np.uint32),
        np.array([30, 20, 10], dtype=np.uint64),
        np.array([300, 200, 100], dtype=np.int64), None,
        np.array([1], dtype=np.uint32), np.array([25], dtype=np.uint64))
    for T, want, vis in [(9, None, False), (10, 100, True), (24, 200, True),
                         (25, None, False), (29, None, False),
                         (30, 300, True), (99, 300, True)]:
        vals, viss = y.oracle_versioned_read(col, T)
        assert (vals[0], viss[0]) == (want, vis), T


def test_versioned_all_empty_rows():
    # rows with no writes and no deletes are never visible
    n = 500
    col = y.encode_versioned_int64(
        np.zeros(n, dtype=np.uint32), np.array([], dtype=np.uint64),
        np.array([], dtype=np.int64), None,
        np.zeros(n, dtype=np.uint32), np.array([], dtype=np.uint64))
    vals, vis = y.oracle_versioned_read(col, 10**18)
    assert vals == [None] * n and vis == [False] * n


def test_versioned_bad_order_errors():
    with pytest.raises(RuntimeError, match="descending"):
        y.encode_versioned_int64(
            np.array([2], dtype=np.uint32), np.array([10, 20], dtype=np.uint64),
            np.array([1, 2], dtype=np.int64), None,
            np.array([0], dtype=np.uint32), np.array([], dtype=np.uint64))


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(3))
def test_versioned_read_gpu_parity(cuda, seed):
    rng = np.random.default_rng([9, seed])
    n = int(rng.choice([97, 20_000]))
    rows, col = _gen(rng, n)
    for T in [0, 100, 500, 999, 10**15]:
        got = y.gpu_versioned_read(col, T, cuda)
        want = y.oracle_versioned_read(col, T)
        assert got == want, f"T={T}"


@pytest.mark.gpu
def test_versioned_read_gpu_multiseg(cuda):
    rng = np.random.default_rng(11)
    rows, _ = _gen(rng, 5000)
    wpr = np.array([len(r[0]) for r in rows], dtype=np.uint32)
    dpr = np.array([len(r[1]) for r in rows], dtype=np.uint32)
    wts = np.array([t for r in rows for t in r[0]], dtype=np.uint64)
    dts = np.array([t for r in rows for t in r[1]], dtype=np.uint64)
    vals = np.array([x for r in rows for x in r[2]], dtype=np.int64)
    vnul = np.array([x for r in rows for x in r[3]], dtype=np.uint8)
    col = y.encode_versioned_int64(wpr, wts, vals, vnul, dpr, dts,
                                   max_rows_per_segment=512)
    for T in [250, 750]:
        assert y.gpu_versioned_read(col, T, cuda) == _model(rows, T)


@pytest.mark.gpu
def test_versioned_scan_chunk_feeds_engine(cuda):
    # the bridge: versioned chunk -> read at T -> compacted DirectDense chunk
    # -> GROUP BY on the engine, vs a pure-Python model of the visible rows
    import collections
    rng = np.random.default_rng(12)
    n = 300_000
    rows, col = _gen(rng, n, ts_space=10_000)
    for T in [2500, 7500]:
        sc = y.gpu_versioned_scan_chunk(col, T)
        vals, vis = _model(rows, T)
        visible_vals = [v for v, s in zip(vals, vis) if s]
        assert sc.row_count == len(visible_vals)
        if sc.row_count == 0:
            continue
        plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum1()])
        got, _ = y.gpu_execute(plan, sc.chunk, max_groups_hint=1 << 19,
                               out_capacity=1 << 19)
        want = collections.Counter(visible_vals)
        assert len(got) == len(want)
        for k, c in got:
            assert want[k] == c
