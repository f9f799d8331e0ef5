"""Encoder (product) ↔ oracle decoder round-trips across all four unversioned
integer segment formats, mirroring the reference's own column round-trip
tests (ytlib/table_client/unittests/integer_column_ut.cpp: CreateDirectDense
:73-84 and the dictionary/RLE variants).

Segment-type selection is checked against the reference's min-estimated-size
rule (integer_column_writer.cpp:353-391,491-535; enum order private.h:25-30).
"""
import ctypes as C

import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd import _abi
from ytsaurus_amd._abi import (
    SEG_DICTIONARY_RLE, SEG_DICTIONARY_DENSE, SEG_DIRECT_RLE, SEG_DIRECT_DENSE,
    YtColumn,
)


def decode_via_oracle(enc, n):
    colc = YtColumn(value_type=enc.value_type,
                    segment_count=enc._cenc.segment_count,
                    segments=enc._cenc.segments)
    vals = np.zeros(max(n, 1), dtype=np.int64)
    nulls = np.zeros(max(n, 1), dtype=np.uint8)
    rc = _abi.oracle_lib().yto_decode_column(
        C.byref(colc), n,
        vals.ctypes.data_as(C.POINTER(C.c_int64)),
        nulls.ctypes.data_as(C.POINTER(C.c_uint8)))
    assert rc == 0
    return vals[:n], nulls[:n]


def check_roundtrip(values, nulls=None, max_seg=0, expect_type=None):
    values = np.asarray(values, dtype=np.int64)
    n = len(values)
    enc = y.encode_int64(values, nulls, max_segment_values=max_seg)
    if expect_type is not None:
        assert enc.segments[0].type == expect_type, enc.segments[0].type
    got_v, got_n = decode_via_oracle(enc, n)
    if nulls is None:
        nulls = np.zeros(n, dtype=np.uint8)
    np.testing.assert_array_equal(got_n, np.asarray(nulls, dtype=np.uint8))
    mask = np.asarray(nulls) == 0
    np.testing.assert_array_equal(got_v[mask], values[mask])
    return enc


def test_direct_dense():
    # integer_column_ut.cpp CreateDirectDense: distinct increasing values
    rng = np.random.default_rng(1)
    vals = rng.integers(-2**40, 2**40, 5000, dtype=np.int64)
    check_roundtrip(vals, expect_type=SEG_DIRECT_DENSE)


def test_direct_dense_with_nulls():
    rng = np.random.default_rng(2)
    vals = rng.integers(-1000, 1000, 3000, dtype=np.int64)
    nulls = (rng.random(3000) < 0.3).astype(np.uint8)
    check_roundtrip(vals, nulls)


def test_dictionary_dense():
    # few distinct wide values, non-repeating pattern → DictionaryDense
    rng = np.random.default_rng(3)
    dict_vals = rng.integers(-2**60, 2**60, 7, dtype=np.int64)
    vals = dict_vals[rng.integers(0, 7, 4096)]
    enc = check_roundtrip(vals, expect_type=SEG_DICTIONARY_DENSE)
    assert enc.segments[0].type == SEG_DICTIONARY_DENSE


def test_direct_rle():
    # long runs of wide distinct values → DirectRle
    vals = np.repeat(
        np.random.default_rng(4).integers(-2**60, 2**60, 40, dtype=np.int64), 100)
    check_roundtrip(vals, expect_type=SEG_DIRECT_RLE)


def test_dictionary_rle():
    # long runs over a tiny dictionary → DictionaryRle
    base = np.array([-2**60, 2**60 - 1, 5], dtype=np.int64)
    vals = base[np.tile([0, 1, 2], 50).repeat(100)]
    check_roundtrip(vals, expect_type=SEG_DICTIONARY_RLE)


def test_rle_with_nulls():
    vals = np.repeat(np.arange(20, dtype=np.int64) * 2**50, 64)
    nulls = np.zeros(len(vals), dtype=np.uint8)
    nulls[640:1280] = 1
    check_roundtrip(vals, nulls)


def test_all_null_segment():
    # MinValue stays 2^64-1, span wraps to 1, width 1 (writer statistics
    # semantics, integer_column_writer.cpp:44-66)
    n = 100
    check_roundtrip(np.zeros(n, dtype=np.int64), np.ones(n, dtype=np.uint8))


def test_single_value():
    check_roundtrip(np.array([42], dtype=np.int64))
    check_roundtrip(np.array([-2**63], dtype=np.int64))
    check_roundtrip(np.array([2**63 - 1], dtype=np.int64))


def test_multi_segment_split():
    # segments split every max_segment_values rows
    rng = np.random.default_rng(5)
    vals = rng.integers(-10**6, 10**6, 2500, dtype=np.int64)
    enc = check_roundtrip(vals, max_seg=1000)
    assert enc._cenc.segment_count == 3
    assert [s.row_count for s in enc.segments] == [1000, 1000, 500]


def test_extreme_values():
    vals = np.array([-2**63, 2**63 - 1, 0, -1, 1], dtype=np.int64)
    check_roundtrip(vals)


def test_empty_column():
    enc = y.encode_int64(np.array([], dtype=np.int64))
    assert enc._cenc.segment_count == 0


def test_double_roundtrip():
    rng = np.random.default_rng(6)
    vals = rng.random(2000)
    nulls = (rng.random(2000) < 0.1).astype(np.uint8)
    enc = y.encode_double(vals, nulls)
    colc = YtColumn(value_type=enc.value_type,
                    segment_count=enc._cenc.segment_count,
                    segments=enc._cenc.segments)
    got = np.zeros(2000, dtype=np.int64)
    gn = np.zeros(2000, dtype=np.uint8)
    rc = _abi.oracle_lib().yto_decode_column(
        C.byref(colc), 2000,
        got.ctypes.data_as(C.POINTER(C.c_int64)),
        gn.ctypes.data_as(C.POINTER(C.c_uint8)))
    assert rc == 0
    np.testing.assert_array_equal(gn, nulls)
    np.testing.assert_array_equal(got.view(np.float64)[nulls == 0], vals[nulls == 0])


def test_boolean_roundtrip():
    """boolean segments — boolean_column_writer.cpp DumpBooleanValues:
    [u64 count][value bitmap][null bitmap], both 8-aligned."""
    rng = np.random.default_rng(7)
    n = 3000
    vals = (rng.random(n) < 0.5).astype(np.uint8)
    nulls = (rng.random(n) < 0.1).astype(np.uint8)
    enc = y.encode_bool(vals, nulls, max_segment_values=1000)
    assert enc._cenc.segment_count == 3
    colc = YtColumn(value_type=enc.value_type,
                    segment_count=enc._cenc.segment_count,
                    segments=enc._cenc.segments)
    got = np.zeros(n, dtype=np.int64)
    gn = np.zeros(n, dtype=np.uint8)
    rc = _abi.oracle_lib().yto_decode_column(
        C.byref(colc), n,
        got.ctypes.data_as(C.POINTER(C.c_int64)),
        gn.ctypes.data_as(C.POINTER(C.c_uint8)))
    assert rc == 0
    np.testing.assert_array_equal(gn, nulls)
    np.testing.assert_array_equal(got[nulls == 0], vals[nulls == 0])


def test_encoder_fuzz_roundtrip():
    """Seeded sweep over widths 1..64, null densities, run/dict shapes and
    segment splits — every encode must decode back exactly (the writer's
    min-size rule picks whatever format it likes; the decode contract is
    format-independent)."""
    rng = np.random.default_rng(99)
    for case in range(30):
        n = int(rng.choice([1, 63, 64, 1000, 5000]))
        bits = int(rng.integers(1, 63))
        style = rng.integers(0, 4)
        if style == 0:
            vals = rng.integers(-(1 << bits), 1 << bits, n, dtype=np.int64)
        elif style == 1:                     # few distinct (dictionary)
            pool = rng.integers(-(1 << bits), 1 << bits, 5, dtype=np.int64)
            vals = pool[rng.integers(0, 5, n)]
        elif style == 2:                     # runs (RLE)
            reps = max(n // 20, 1)
            vals = np.repeat(
                rng.integers(-(1 << bits), 1 << bits, reps, dtype=np.int64),
                20)[:n]
            if len(vals) < n:
                vals = np.pad(vals, (0, n - len(vals)), constant_values=3)
        else:                                # constant
            vals = np.full(n, int(rng.integers(-(1 << bits), 1 << bits)),
                           dtype=np.int64)
        nulls = None
        if rng.random() < 0.6:
            nulls = (rng.random(n) < rng.choice([0.0, 0.05, 0.5, 1.0])
                     ).astype(np.uint8)
        seg = int(rng.choice([0, 100, 1024]))
        check_roundtrip(vals, nulls, max_seg=seg)
