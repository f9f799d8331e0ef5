"""Known-answer tests for the bit-packed unsigned vector format.

The expected words below are hand-computed from the reference algorithm
(core/misc/bit_packed_unsigned_vector-inl.h:32-82):
header = count (low 56 bits) | width (top 8 bits), width = MSB(maxValue)+1;
values packed LSB-first, straddling word boundaries; width-64 raw words;
maxValue == 0 → header only.

These pin the PRODUCT encoder (yt_bitpack) and the ORACLE decoder
(yto_bitunpack) independently against the reference layout, so a symmetric
bug cannot survive round-trip testing.
"""
import ctypes as C

import numpy as np
import pytest

from ytsaurus_amd import _abi


def pack(values, maxv):
    lib = _abi.gpu_lib()
    n = len(values)
    words = lib.yt_bitpack_size_words(maxv, n)
    buf = (C.c_uint64 * words)()
    arr = (C.c_uint64 * n)(*values)
    used = lib.yt_bitpack(arr, n, maxv, buf)
    assert used == words, (used, words)
    return [buf[i] for i in range(words)]


def unpack(words):
    lib = _abi.oracle_lib()
    buf = (C.c_uint64 * len(words))(*words)
    out = (C.c_uint64 * (1 << 16))()
    n = lib.yto_bitunpack(buf, out, 1 << 16)
    return [out[i] for i in range(n)]


def test_header_word():
    # width(3) = 2 → header = 3 | 2<<56
    w = pack([1, 2, 3], 3)
    assert w[0] == 3 | (2 << 56)
    # values 1,2,3 at 2 bits LSB-first: 0b11_10_01 = 0x39
    assert w[1] == 0b111001
    assert unpack(w) == [1, 2, 3]


def test_width_zero():
    # maxValue 0 → header only, width 0; reader returns zeros
    w = pack([0, 0, 0, 0], 0)
    assert len(w) == 1
    assert w[0] == 4  # count 4, width 0
    assert unpack(w) == [0, 0, 0, 0]


def test_word_straddle():
    # width 7, 10 values: 70 bits → 2 data words; value 9 straddles bit 63/64
    vals = [(i * 11 + 3) % 128 for i in range(10)]
    w = pack(vals, 127)
    assert w[0] == 10 | (7 << 56)
    # recompute expected by direct bit placement
    exp = [0, 0]
    for i, v in enumerate(vals):
        bit = i * 7
        exp[bit // 64] |= (v << (bit % 64)) & ((1 << 64) - 1)
        if bit % 64 + 7 > 64:
            exp[bit // 64 + 1] |= v >> (64 - bit % 64)
    assert w[1:] == exp
    assert unpack(w) == vals


def test_width_64_raw():
    vals = [2**64 - 1, 0, 123456789, 2**63]
    w = pack(vals, 2**64 - 1)
    assert w[0] == 4 | (64 << 56)
    assert w[1:] == vals
    assert unpack(w) == vals


@pytest.mark.parametrize("width", list(range(1, 65)))
def test_roundtrip_every_width(width):
    rng = np.random.default_rng(width)
    maxv = (1 << width) - 1
    vals = [int(x) for x in
            rng.integers(0, 2**width, 257, dtype=np.uint64, endpoint=False)]
    vals[0] = maxv  # force the full width
    w = pack(vals, maxv)
    assert w[0] == 257 | (width << 56)
    assert unpack(w) == vals
