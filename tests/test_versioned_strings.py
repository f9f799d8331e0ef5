"""Versioned STRING value columns (§8f row 3 completion): direct and
dictionary string layouts under the dense/sparse value index + aggregate
bitmap (string_column_writer.cpp TVersionedStringColumnWriter::DumpSegment
:325-360, DumpDirectValues :205-229, DumpDictionaryValues :153-203; the
versioned Any/Composite writers share this byte layout :365-401). Parity
pinned against an independent pure-Python model of the rowset_builder
visibility rule (rowset_builder.cpp:1042-1166)."""
import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd import _abi


def model(rows, T):
    vals, vis, agg = [], [], []
    for w, d, v, nl, ag in rows:
        delete_ts = next((ts for ts in d if ts <= T), 0)
        visible = [(i, ts) for i, ts in enumerate(w) if delete_ts < ts <= T]
        if not visible:
            vals.append(None)
            vis.append(False)
            agg.append(False)
            continue
        i, _ = visible[0]
        vals.append(None if nl[i] else v[i])
        vis.append(True)
        agg.append(bool(ag[i]))
    return vals, vis, agg


def flatten(rows):
    wpr = np.array([len(r[0]) for r in rows], dtype=np.uint32)
    dpr = np.array([len(r[1]) for r in rows], dtype=np.uint32)
    wts = np.array([t for r in rows for t in r[0]], dtype=np.uint64)
    vals = [x for r in rows for x in r[2]]
    nuls = np.array([x for r in rows for x in r[3]], dtype=np.uint8)
    dts = np.array([t for r in rows for t in r[1]], dtype=np.uint64)
    aggs = np.array([x for r in rows for x in r[4]], dtype=np.uint8)
    return wpr, wts, vals, nuls, dpr, dts, aggs


def gen_rows(rng, n, pool=None, max_writes=3, sparse_shape=False,
             varlen=True):
    rows = []
    for _ in range(n):
        if sparse_shape:
            wc = int(rng.integers(0, 8)) if rng.random() < 0.03 else 0
        else:
            wc = int(rng.integers(0, max_writes + 1))
        dc = int(rng.integers(0, 2))
        w = (np.sort(rng.choice(1000, size=wc, replace=False))[::-1] + 1).tolist()
        d = (np.sort(rng.choice(1000, size=dc, replace=False))[::-1] + 1).tolist()
        nl = (rng.random(wc) < 0.15).astype(np.uint8).tolist()
        ag = (rng.random(wc) < 0.5).astype(np.uint8).tolist()
        v = []
        for i in range(wc):
            if nl[i]:
                v.append(b"")
                continue
            if pool is not None:
                v.append(pool[int(rng.integers(0, len(pool)))])
            else:
                ln = int(rng.integers(0, 24)) if varlen else 10
                v.append(bytes(rng.integers(97, 123, ln, dtype=np.uint8)))
        rows.append((w, d, v, nl, ag))
    return rows


TS = [0, 1, 250, 500, 999, 1000, 10**15]


def norm(v, nl):
    return None if nl else v


def check_oracle(col, rows, with_agg):
    for T in TS:
        want_v, want_vis, want_agg = model(rows, T)
        want_v = [None if x is None else bytes(x) for x in want_v]
        if with_agg:
            got_v, got_vis, got_agg = y.oracle_versioned_read(col, T,
                                                              with_agg=True)
            assert got_agg == want_agg, f"T={T}"
        else:
            got_v, got_vis = y.oracle_versioned_read(col, T)
        assert got_vis == want_vis, f"T={T}"
        assert got_v == want_v, f"T={T}"


def test_versioned_string_direct():
    rng = np.random.default_rng(201)
    rows = gen_rows(rng, 8_000)          # mostly-unique strings -> direct
    wpr, wts, vals, nuls, dpr, dts, aggs = flatten(rows)
    col = y.encode_versioned_string(wpr, wts, vals, nuls, dpr, dts)
    types = {col._c.val_segs[i].type for i in range(col._c.val_seg_count)}
    assert all(t >= 32 for t in types)
    assert any(t in (32, 34) for t in types), f"expected direct, got {types}"
    check_oracle(col, rows, with_agg=False)


def test_versioned_string_dictionary():
    rng = np.random.default_rng(202)
    pool = [b"alpha", b"beta", b"gamma-longish-value"]
    rows = gen_rows(rng, 8_000, pool=pool, max_writes=4)
    wpr, wts, vals, nuls, dpr, dts, aggs = flatten(rows)
    col = y.encode_versioned_string(wpr, wts, vals, nuls, dpr, dts)
    types = {col._c.val_segs[i].type for i in range(col._c.val_seg_count)}
    assert any(t in (33, 35) for t in types), f"expected dict, got {types}"
    check_oracle(col, rows, with_agg=False)


def test_versioned_string_sparse_agg():
    rng = np.random.default_rng(203)
    rows = gen_rows(rng, 25_000, sparse_shape=True)
    wpr, wts, vals, nuls, dpr, dts, aggs = flatten(rows)
    col = y.encode_versioned_string(wpr, wts, vals, nuls, dpr, dts,
                                    value_agg=aggs)
    types = {col._c.val_segs[i].type for i in range(col._c.val_seg_count)}
    assert any(t & 2 for t in types), f"expected sparse, got {types}"
    assert all(col._c.val_segs[i].flags & _abi.VSEG_F_AGGREGATE
               for i in range(col._c.val_seg_count))
    check_oracle(col, rows, with_agg=True)


def test_versioned_string_multiseg_and_empty():
    rng = np.random.default_rng(204)
    pool = [b"", b"x", b"same", b"same", b"other"]
    rows = gen_rows(rng, 12_000, pool=pool)
    wpr, wts, vals, nuls, dpr, dts, aggs = flatten(rows)
    col = y.encode_versioned_string(wpr, wts, vals, nuls, dpr, dts,
                                    max_rows_per_segment=4096)
    assert col._c.val_seg_count == 3
    check_oracle(col, rows, with_agg=False)


def test_versioned_string_null_with_bytes_rejected():
    with pytest.raises(ValueError):
        # lens mismatch with writes
        y.encode_versioned_string(np.array([1], dtype=np.uint32),
                                  np.array([5], dtype=np.uint64),
                                  [], None,
                                  np.array([0], dtype=np.uint32),
                                  np.array([], dtype=np.uint64))


@pytest.mark.gpu
@pytest.mark.parametrize("kind", ["direct", "dict", "sparse", "agg"])
def test_versioned_string_gpu_parity(cuda, kind):
    rng = np.random.default_rng([205, hash(kind) % 1000])
    if kind == "direct":
        rows = gen_rows(rng, 10_000)
    elif kind == "dict":
        rows = gen_rows(rng, 8_000, pool=[b"aa", b"bb", b"cc-long-value"])
    elif kind == "sparse":
        rows = gen_rows(rng, 25_000, sparse_shape=True)
    else:
        rows = gen_rows(rng, 10_000)
    wpr, wts, vals, nuls, dpr, dts, aggs = flatten(rows)
    va = aggs if kind == "agg" else None
    col = y.encode_versioned_string(wpr, wts, vals, nuls, dpr, dts,
                                    value_agg=va,
                                    max_rows_per_segment=4096)
    for T in [250, 750, 1000]:
        want = y.oracle_versioned_read(col, T, with_agg=va is not None)
        got = y.gpu_versioned_read(col, T, cuda, with_agg=va is not None)
        assert got == want, f"T={T}"


@pytest.mark.gpu
def test_versioned_string_bridge_refused(cuda):
    rng = np.random.default_rng(206)
    rows = gen_rows(rng, 1_000)
    wpr, wts, vals, nuls, dpr, dts, _ = flatten(rows)
    col = y.encode_versioned_string(wpr, wts, vals, nuls, dpr, dts)
    with pytest.raises(RuntimeError, match="not yet bridged"):
        y.gpu_versioned_scan_chunk(col, 500)
