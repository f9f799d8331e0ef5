"""Versioned scan-format fuzz (CPU): random MVCC columns — int64, double
and string values, random write/delete densities and segment sizes — read
at a sweep of timestamps by the ORACLE against the independent pure-Python
visibility model (rowset_builder.cpp:1042-1166). Widens the directed
shapes in test_versioned2/test_versioned_strings to the random corner
space (empty rows, all-deleted rows, dictionary-vs-direct flips at the
segment level, sparse/dense index flips)."""
import numpy as np
import pytest

import ytsaurus_amd as y
from test_versioned2 import model, flatten


def gen(rng, n, kind):
    rows = []
    sparse = rng.random() < 0.4
    pool = None
    if kind == "int" and rng.random() < 0.5:
        pool = rng.integers(-50, 50, int(rng.integers(2, 6))).tolist()
    if kind == "str":
        pool = ([b"x", b"yy", b"zzz", b""] if rng.random() < 0.5 else None)
    for _ in range(n):
        if sparse:
            wc = int(rng.integers(0, 6)) if rng.random() < 0.05 else 0
        else:
            wc = int(rng.integers(0, 4))
        dc = int(rng.integers(0, 3))
        w = (np.sort(rng.choice(2000, size=wc, replace=False))[::-1] + 1).tolist()
        d = (np.sort(rng.choice(2000, size=dc, replace=False))[::-1] + 1).tolist()
        nl = (rng.random(wc) < 0.2).astype(np.uint8).tolist()
        if kind == "int":
            if pool is not None:
                v = [int(pool[int(i)]) for i in rng.integers(0, len(pool), wc)]
            else:
                v = rng.integers(-10**12, 10**12, wc).tolist()
        elif kind == "double":
            v = (rng.random(wc) * 1e6 - 5e5).tolist()
        else:
            v = []
            for i in range(wc):
                if nl[i]:
                    v.append(b"")
                elif pool is not None:
                    v.append(pool[int(rng.integers(0, len(pool)))])
                else:
                    ln = int(rng.integers(0, 20))
                    v.append(bytes(rng.integers(97, 123, ln, dtype=np.uint8)))
        ag = (rng.random(wc) < 0.5).astype(np.uint8).tolist()
        rows.append((w, d, v, nl, ag))
    return rows


@pytest.mark.parametrize("seed", range(24))
def test_versioned_oracle_fuzz(seed):
    rng = np.random.default_rng([20260917, seed])
    kind = ["int", "double", "str"][seed % 3]
    n = int(rng.choice([37, 3000, 12000]))
    rows = gen(rng, n, kind)
    wpr, wts, vals, nuls, dpr, dts, aggs = flatten(rows)
    with_agg = bool(rng.integers(0, 2))
    seg = int(rng.choice([0, 512, 4096]))
    va = aggs if with_agg else None
    if kind == "int":
        col = y.encode_versioned_int64(wpr, wts, np.array(vals, dtype=np.int64),
                                       nuls, dpr, dts, value_agg=va,
                                       max_rows_per_segment=seg)
    elif kind == "double":
        col = y.encode_versioned_double(wpr, wts,
                                        np.array(vals, dtype=np.float64),
                                        nuls, dpr, dts, value_agg=va,
                                        max_rows_per_segment=seg)
    else:
        col = y.encode_versioned_string(wpr, wts, vals, nuls, dpr, dts,
                                        value_agg=va,
                                        max_rows_per_segment=seg)
    for T in [0, 1, int(rng.integers(2, 2000)), 1000, 2001, 10**15]:
        want_v, want_vis, want_agg = model(rows, T)
        if kind == "str":
            want_v = [None if x is None else bytes(x) for x in want_v]
        if with_agg:
            got_v, got_vis, got_agg = y.oracle_versioned_read(col, T,
                                                              with_agg=True)
            assert got_agg == want_agg, (seed, T)
        else:
            got_v, got_vis = y.oracle_versioned_read(col, T)
        assert got_vis == want_vis, (seed, T)
        for a, b in zip(got_v, want_v):
            if isinstance(b, float):
                assert a == b, (seed, T)
            else:
                assert a == b, (seed, T)
