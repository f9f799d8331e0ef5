"""GPU parity: the product HIP path vs the CPU oracle on identical seeded
inputs (small sizes, exact rowset compare after sorting by key — the
reference's own OrderedResultMatcher discipline,
unittests/evaluate/test_evaluate.cpp:174-199), plus size-independent
properties at larger sizes. All tests @pytest.mark.gpu."""
import ctypes as C

import numpy as np
import pytest

import ytsaurus_amd as y
from helpers import build_chunk, build_plan, load_cases, norm_rows

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")


def run_both(plan, chunk, cuda, hint=1 << 16):
    dev = chunk.c_device(cuda)
    got, stats = y.gpu_execute(plan, dev, max_groups_hint=hint)
    want, _ = y.oracle_execute(plan, chunk)
    return got, want, stats


CASES = [c for c in load_cases() if c["plan"].get("aggs")]


@pytest.mark.parametrize("case", CASES, ids=[c["name"] for c in CASES])
def test_golden_on_gpu(case, cuda):
    chunk = build_chunk(case["columns"], case["rows"])
    plan = build_plan(case["plan"])
    got, want, _ = run_both(plan, chunk, cuda, hint=1 << 12)
    assert y.sort_rows(norm_rows(got)) == y.sort_rows(norm_rows(want))


def _mk(rng, n, key_range, null_frac=0.0, seg=0):
    keys = rng.integers(-key_range, key_range, n, dtype=np.int64)
    vals = rng.integers(-10**9, 10**9, n, dtype=np.int64)
    kn = (rng.random(n) < null_frac).astype(np.uint8) if null_frac else None
    vn = (rng.random(n) < null_frac).astype(np.uint8) if null_frac else None
    return y.Chunk([y.encode_int64(keys, kn, max_segment_values=seg),
                    y.encode_int64(vals, vn, max_segment_values=seg)], n)


def group_plan(filtered=False):
    f = (y.col(0) > 0) if filtered else None
    return y.Plan(filter=f, keys=[y.col(0)],
                  aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])


@pytest.mark.parametrize("n,kr,nf", [
    (1000, 50, 0.0),        # small, few groups
    (100_000, 1000, 0.0),   # fast path, multi-tile
    (100_000, 1000, 0.1),   # nulls in keys and values
    (300_000, 100_000, 0.0),  # many groups, multiple segments
    (1000, 2**62, 0.0),     # near-distinct wide keys
])
def test_group_by_parity(cuda, n, kr, nf):
    rng = np.random.default_rng(n + kr % 997)
    chunk = _mk(rng, n, kr, nf)
    hint = min(max(4 * kr, 1 << 12), 1 << 21)
    got, want, stats = run_both(group_plan(), chunk, cuda, hint=hint)
    assert y.sort_rows(got) == y.sort_rows(want)
    assert stats.kernel_scan_launches >= 1


def test_group_by_filter_parity(cuda):
    rng = np.random.default_rng(42)
    chunk = _mk(rng, 200_000, 500, 0.05)
    got, want, _ = run_both(group_plan(filtered=True), chunk, cuda, hint=1 << 12)
    assert y.sort_rows(got) == y.sort_rows(want)


def test_zero_and_min_keys(cuda):
    """key 0 (the table's EMPTY sentinel) and INT64_MIN via side slots."""
    keys = np.array([0, 0, -2**63, 1, 0, -2**63, 2**63 - 1], dtype=np.int64)
    vals = np.arange(7, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(keys), y.encode_int64(vals)], 7)
    got, want, _ = run_both(group_plan(), chunk, cuda, hint=64)
    assert y.sort_rows(got) == y.sort_rows(want)


def test_null_keys(cuda):
    rng = np.random.default_rng(3)
    n = 50_000
    keys = rng.integers(0, 100, n, dtype=np.int64)
    kn = (rng.random(n) < 0.2).astype(np.uint8)
    vals = rng.integers(-5, 5, n, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(keys, kn), y.encode_int64(vals)], n)
    got, want, _ = run_both(group_plan(), chunk, cuda, hint=1 << 10)
    assert y.sort_rows(got) == y.sort_rows(want)


def test_generic_expressions(cuda):
    """expression key + expression agg → generic kernel path"""
    rng = np.random.default_rng(4)
    n = 100_000
    chunk = _mk(rng, n, 10**6, 0.05)
    plan = y.Plan(filter=(y.col(0) % 7) != 0,
                  keys=[(y.col(0) % 101) * 2 + 1],
                  aggs=[y.agg_sum(y.col(1) % 1000), y.agg_sum1()])
    got, want, _ = run_both(plan, chunk, cuda, hint=1 << 12)
    assert y.sort_rows(got) == y.sort_rows(want)


@pytest.mark.parametrize("shape", ["dict_dense", "direct_rle", "dict_rle"])
def test_non_dense_segments(cuda, shape):
    rng = np.random.default_rng(hash(shape) % 2**31)
    n = 100_000
    if shape == "dict_dense":
        dict_vals = rng.integers(-2**60, 2**60, 17, dtype=np.int64)
        keys = dict_vals[rng.integers(0, 17, n)]
    elif shape == "direct_rle":
        keys = np.repeat(rng.integers(-2**60, 2**60, n // 100, dtype=np.int64), 100)
    else:
        base = rng.integers(-2**60, 2**60, 5, dtype=np.int64)
        keys = base[np.tile(np.arange(5), n // 500).repeat(100)]
    vals = rng.integers(-10**9, 10**9, len(keys), dtype=np.int64)
    kn = (rng.random(len(keys)) < 0.02).astype(np.uint8)
    chunk = y.Chunk([y.encode_int64(keys, kn), y.encode_int64(vals)], len(keys))
    got, want, _ = run_both(group_plan(), chunk, cuda, hint=1 << 12)
    assert y.sort_rows(got) == y.sort_rows(want)


def test_global_aggregate_fast(cuda):
    """BASELINE config-2 shape: range filter + several sums, no keys."""
    rng = np.random.default_rng(5)
    n = 1_000_000
    c0 = rng.integers(0, 2**40, n, dtype=np.int64)
    c1 = rng.integers(-10**9, 10**9, n, dtype=np.int64)
    c2 = rng.integers(-10**9, 10**9, n, dtype=np.int64)
    c3 = rng.integers(-10**9, 10**9, n, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(c) for c in (c0, c1, c2, c3)], n)
    lo, hi = 2**38, 3 * 2**38
    plan = y.Plan(filter=(y.col(0) >= lo).and_(y.col(0) <= hi),
                  aggs=[y.agg_sum(y.col(1)), y.agg_sum(y.col(2)),
                        y.agg_sum(y.col(3)), y.agg_sum1()])
    got, want, _ = run_both(plan, chunk, cuda)
    assert got == want
    # cross-check against numpy on the same inputs
    mask = (c0 >= lo) & (c0 <= hi)
    assert got[0][0] == int(c1[mask].sum())
    assert got[0][3] == int(mask.sum())


def test_empty_and_allnull(cuda):
    empty = y.Chunk([y.encode_int64(np.array([], dtype=np.int64)),
                     y.encode_int64(np.array([], dtype=np.int64))], 0)
    got, want, _ = run_both(group_plan(), empty, cuda, hint=64)
    assert got == want == []

    n = 1000
    chunk = y.Chunk([y.encode_int64(np.zeros(n, dtype=np.int64), np.ones(n, dtype=np.uint8)),
                     y.encode_int64(np.zeros(n, dtype=np.int64), np.ones(n, dtype=np.uint8))], n)
    got, want, _ = run_both(group_plan(), chunk, cuda, hint=64)
    assert y.sort_rows(got) == y.sort_rows(want)


def test_two_phase_gpu_matches_single(cuda):
    """partial(8 partitions) + merge == single-pass execute; and the state
    rows agree with the CPU oracle's partials."""
    rng = np.random.default_rng(6)
    n = 500_000
    chunk = _mk(rng, n, 10_000, 0.02)
    plan = group_plan()

    single, stats = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=1 << 16)

    nparts = 8
    cap = 4 * 10_000 + 64
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    counts, _ = y.gpu_partial(plan, chunk.c_device(cuda), nparts,
                              states_t.data_ptr(), cap, max_groups_hint=1 << 16)
    total = sum(counts)

    merged = []
    at = 0
    for p in range(nparts):
        if counts[p] == 0:
            continue
        part = states_t[at:at + counts[p]].contiguous()
        rows, _ = y.gpu_merge(plan, part.data_ptr(), counts[p],
                              max_groups_hint=1 << 16)
        merged += rows
        at += counts[p]

    assert y.sort_rows(merged) == y.sort_rows(single)

    # property checks at size: sums of sums and counts match a global run
    # (single rows are [key, sum(v), sum(1)])
    glob = y.Plan(aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])
    grow, _ = y.gpu_execute(glob, chunk.c_device(cuda))
    tot_cnt = sum(r[2] for r in single)
    assert tot_cnt == grow[0][1] == n
    M = 1 << 64
    tot_sum = sum((r[1] if r[1] is not None else 0) % M for r in single) % M
    want_sum = (grow[0][0] or 0) % M
    assert tot_sum == want_sum
    assert total == len(single)


def test_group_row_limit_incomplete(cuda):
    """GroupRowLimit semantics: processing is interrupted and the output is
    flagged incomplete (registry.cpp:1490 + 1902-1907). Out-of-order GPU
    processing makes the retained subset unspecified; the FLAG must match."""
    rng = np.random.default_rng(7)
    chunk = _mk(rng, 10_000, 5000, 0.0)
    plan = group_plan()
    dev = chunk.c_device(cuda)
    got, stats = y.gpu_execute(plan, dev, max_groups_hint=1 << 16,
                               group_row_limit=10)
    assert stats.incomplete_output == 1


def test_double_sum_parity(cuda):
    """sum over a DOUBLE column (generic kernel, fp64 atomics): within 1e-6
    relative of the oracle — the bound the reference's own nondeterministic
    cross-tablet merge order implies (north_star; test_evaluate.cpp:220-260
    checks doubles at 1e-5 absolute)."""
    rng = np.random.default_rng(11)
    n = 200_000
    keys = rng.integers(0, 500, n, dtype=np.int64)
    vals = rng.random(n)
    vn = (rng.random(n) < 0.05).astype(np.uint8)
    chunk = y.Chunk([y.encode_int64(keys), y.encode_double(vals, vn)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])
    got, stats = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=2048)
    want, _ = y.oracle_execute(plan, chunk)
    gm = {r[0]: r for r in got}
    assert len(got) == len(want)
    for k, ws, wc in want:
        gk, gs, gc = gm[k]
        assert gc == wc
        if ws is None:
            assert gs is None
        else:
            assert abs(gs - ws) <= 1e-6 * max(abs(ws), 1e-30)


def test_statistics_counters(cuda):
    """TQueryStatistics parity: RowsRead / RowsWritten / GroupedRowCount
    (query_statistics.h:49-79 semantics)."""
    rng = np.random.default_rng(12)
    n = 100_000
    chunk = _mk(rng, n, 100, 0.0)
    got, stats = y.gpu_execute(group_plan(), chunk.c_device(cuda), max_groups_hint=1024)
    assert stats.rows_read == n
    assert stats.rows_written == len(got)
    assert stats.grouped_row_count == len(got)
    assert stats.data_weight_read > 0


def test_min_max_parity(cuda):
    """min/max aggregates (udf/min.c, max.c semantics) on the generic path,
    int64 and double, with nulls."""
    from ytsaurus_amd._abi import AGG_MIN, AGG_MAX
    rng = np.random.default_rng(13)
    n = 100_000
    keys = rng.integers(0, 300, n, dtype=np.int64)
    iv = rng.integers(-2**62, 2**62, n, dtype=np.int64)
    dv = rng.standard_normal(n) * 1e6
    ivn = (rng.random(n) < 0.1).astype(np.uint8)
    chunk = y.Chunk([y.encode_int64(keys), y.encode_int64(iv, ivn),
                     y.encode_double(dv)], n)
    plan = y.Plan(keys=[y.col(0)],
                  aggs=[(AGG_MIN, y.col(1)), (AGG_MAX, y.col(1)),
                        (AGG_MIN, y.col(2)), (AGG_MAX, y.col(2))])
    got, want, _ = run_both(plan, chunk, cuda, hint=1024)
    assert y.sort_rows(got) == y.sort_rows(want)


SCAN_CASES = [c for c in load_cases() if not c["plan"].get("aggs")]


@pytest.mark.parametrize("case", SCAN_CASES, ids=[c["name"] for c in SCAN_CASES])
def test_scan_project_goldens_on_gpu(case, cuda):
    """scan+filter+project goldens (SimpleWithNull etc.) — order-preserving,
    compared EXACTLY like the reference's ResultMatcher."""
    chunk = build_chunk(case["columns"], case["rows"])
    plan = build_plan(case["plan"])
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda))
    want = norm_rows([tuple(r) for r in case["expected"]])
    assert norm_rows(got) == want


def test_scan_project_large(cuda):
    rng = np.random.default_rng(14)
    n = 500_000
    a = rng.integers(0, 1000, n, dtype=np.int64)
    b = rng.integers(-10**9, 10**9, n, dtype=np.int64)
    bn = (rng.random(n) < 0.05).astype(np.uint8)
    chunk = y.Chunk([y.encode_int64(a), y.encode_int64(b, bn)], n)
    plan = y.Plan(filter=y.col(0) < 100,
                  projects=[y.col(0), y.col(1) + y.col(0), y.col(1) % 7])
    got, st = y.gpu_execute(plan, chunk.c_device(cuda),
                            out_capacity=n + 16)
    want, _ = y.oracle_execute(plan, chunk)
    assert got == want         # exact, order-preserving
    assert st.rows_written == len(want)


def test_input_output_row_limits(cuda):
    """TExecutionContext InputRowLimit / OutputRowLimit semantics
    (registry.cpp:259-265, :297-305): first-N-rows scan with
    incomplete_input, output soft-stop with incomplete_output."""
    import ctypes as C
    from ytsaurus_amd import _abi
    from ytsaurus_amd.api import _mk_rowset
    rng = np.random.default_rng(15)
    n = 300_000
    keys = rng.integers(0, 1000, n, dtype=np.int64)
    vals = rng.integers(0, 100, n, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(keys), y.encode_int64(vals)], n)
    plan = group_plan()
    dev = chunk.c_device(cuda)

    # input limit: equals a full run over the first 200k rows
    opts = _abi.YtExecOptions(input_row_limit=200_000, max_groups_hint=4096)
    rs = _mk_rowset(1 << 16)
    st = _abi.YtStatistics()
    err = C.create_string_buffer(512)
    rc = _abi.gpu_lib().yt_gpu_query_execute(
        C.byref(plan.c), C.byref(dev), C.byref(opts), C.byref(rs), C.byref(st), err, 512)
    assert rc == 0, err.value
    assert st.incomplete_input == 1 and st.rows_read == 200_000
    got = y.rows_from_rowset(rs)
    sub = y.Chunk([y.encode_int64(keys[:200_000]), y.encode_int64(vals[:200_000])], 200_000)
    want, _ = y.oracle_execute(plan, sub)
    assert y.sort_rows(got) == y.sort_rows(want)

    # output limit: at most 10 rows, flagged incomplete
    opts = _abi.YtExecOptions(output_row_limit=10, max_groups_hint=4096)
    rs = _mk_rowset(1 << 16)
    rc = _abi.gpu_lib().yt_gpu_query_execute(
        C.byref(plan.c), C.byref(dev), C.byref(opts), C.byref(rs), C.byref(st), err, 512)
    assert rc == 0, err.value
    assert rs.row_count == 10
    assert st.incomplete_output == 1


def test_bool_and_uint64_columns(cuda):
    """boolean group keys (boolean column format) and uint64 sums on the
    generic path."""
    rng = np.random.default_rng(16)
    n = 80_000
    bk = (rng.random(n) < 0.4).astype(np.uint8)
    bn = (rng.random(n) < 0.1).astype(np.uint8)
    uv = rng.integers(0, 2**63, n, dtype=np.int64)  # bits used as u64
    chunk = y.Chunk([y.encode_bool(bk, bn), y.encode_int64(uv, unsigned=True)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])
    got, want, _ = run_both(plan, chunk, cuda, hint=64)
    assert y.sort_rows(got) == y.sort_rows(want)


def test_string_group_by_gpu(cuda):
    """config-5 family on the GPU: dictionary-encoded string key + double
    sum + sum(1); exact cross-segment merge (hash gate + byte compare)."""
    rng = np.random.default_rng(17)
    n = 120_000
    keyset = ["key-%05d" % i for i in range(300)]
    kidx = rng.integers(0, 300, n)
    kn = rng.random(n) < 0.02
    keys = [None if kn[i] else keyset[int(kidx[i])] for i in range(n)]
    vals = rng.random(n)
    vn = (rng.random(n) < 0.05).astype(np.uint8)
    chunk = y.Chunk([y.encode_string(keys, max_segment_values=20000),
                     y.encode_double(vals, vn, max_segment_values=20000)], n)
    # ensure dictionary encoding was chosen (the GPU path requires it):
    # string segment types are 1 = DictionaryDense, 0 = DictionaryRle
    assert all(s.type in (0, 1) for s in chunk.columns[0].segments)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])
    got, st = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=1024)
    want, _ = y.oracle_execute(plan, chunk)
    gm = {r[0]: r for r in got}
    assert len(got) == len(want)
    for k, sv, cv in want:
        gk, gs, gc = gm[k]
        assert gc == cv
        if sv is None:
            assert gs is None
        else:
            assert abs(gs - sv) <= 1e-6 * max(abs(sv), 1e-30)


def test_string_group_by_gpu_int_sum(cuda):
    """string key + int64 sum: bit-exact."""
    rng = np.random.default_rng(18)
    n = 60_000
    keyset = ["s%d" % i for i in range(500)]
    keys = [keyset[int(i)] for i in rng.integers(0, 500, n)]
    vals = rng.integers(-10**9, 10**9, n, dtype=np.int64)
    chunk = y.Chunk([y.encode_string(keys, max_segment_values=8192),
                     y.encode_int64(vals, max_segment_values=8192)], n)
    plan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])
    got, _ = y.gpu_execute(plan, chunk.c_device(cuda), max_groups_hint=1024)
    want, _ = y.oracle_execute(plan, chunk)
    assert y.sort_rows(got) == y.sort_rows(want)


@pytest.mark.gpu
def test_scan_project_streams_past_16M(cuda):
    """the per-row materialization buffer is windowed (16Mi rows): larger
    inputs stream through bounded memory (a selective filter keeps the
    OUTPUT small while the INPUT spans two windows)"""
    n = 17_000_000
    a = np.arange(n, dtype=np.int64)
    chunk = y.Chunk([y.encode_int64(a)], n)
    plan = y.Plan(filter=(y.col(0) % 1_000_000) == 7,
                  projects=[y.col(0), y.col(0) * 2])
    got, st = y.gpu_execute(plan, chunk.c_device(cuda), out_capacity=64)
    want = [(int(v), int(v) * 2) for v in range(7, n, 1_000_000)]
    assert got == want
    assert st.rows_read == n
