"""High-level Python helpers around the C-ABI (plan building, chunk encoding,
execution). Python here is plumbing only: encoding and all query compute run
in native code (libytql_gpu.so / the oracle).
"""
import ctypes as C

import numpy as np

from . import _abi
from ._abi import (
    YtValue, YtSegment, YtColumn, YtChunk, YtExpr, YtAgg, YtPlan,
    YtExecOptions, YtStatistics, YtRowset, YtStateRow, YtEncodedColumn,
    VT_NULL, VT_INT64, VT_UINT64, VT_DOUBLE, VT_BOOLEAN, VT_STRING,
    EX_COLUMN, EX_LIT_I64, EX_LIT_NULL, EX_LIT_DOUBLE,
    EX_ADD, EX_SUB, EX_MUL, EX_DIV, EX_MOD,
    EX_EQ, EX_NE, EX_LT, EX_LE, EX_GT, EX_GE, EX_AND, EX_OR, EX_NOT,
    AGG_SUM, AGG_SUM1, AGG_MIN, AGG_MAX, AGG_FIRST, AGG_AVG, YT_OK,
)

__all__ = [
    "col", "lit", "null", "litf", "Plan", "Join", "agg_sum", "agg_sum1",
    "agg_first", "agg_avg", "agg_min", "agg_max",
    "encode_int64", "encode_double", "encode_bool", "encode_string", "encode_string_raw",
    "oracle_decode_strings", "encode_versioned_int64", "encode_versioned_double", "encode_versioned_string", "oracle_versioned_read", "gpu_versioned_read", "gpu_versioned_scan_chunk", "gpu_versioned_scan_table", "ScanChunk", "VersionedColumn",
    "Chunk", "oracle_execute",
    "oracle_partial", "oracle_merge", "oracle_partial_mk", "oracle_merge_mk",
    "gpu_key_ranges", "gpu_partial_mk", "gpu_merge_mk",
    "oracle_partial_str", "oracle_merge_str", "gpu_partial_str", "gpu_merge_str",
    "gpu_available", "gpu_execute", "gpu_partial", "gpu_merge",
    "rows_from_rowset", "sort_rows", "make_rowset", "coordinate_results",
]


# ---------------- expression builder ----------------

class Expr:
    """Builds YtExpr trees; keeps ctypes objects alive via _keep."""

    def __init__(self, op, col=-1, lit_i64=0, lit_dbl=0.0, a=None, b=None):
        self.c = YtExpr(op=op, col=col, lit_i64=lit_i64, lit_dbl=lit_dbl)
        self._keep = []
        if a is not None:
            self.c.a = C.pointer(a.c)
            self._keep.append(a)
        if b is not None:
            self.c.b = C.pointer(b.c)
            self._keep.append(b)

    def _bin(self, op, other):
        return Expr(op, a=self, b=_coerce(other))

    def __add__(self, o): return self._bin(EX_ADD, o)
    def __sub__(self, o): return self._bin(EX_SUB, o)
    def __mul__(self, o): return self._bin(EX_MUL, o)
    def __floordiv__(self, o): return self._bin(EX_DIV, o)
    def __mod__(self, o): return self._bin(EX_MOD, o)
    def __eq__(self, o): return self._bin(EX_EQ, o)      # noqa: E704
    def __ne__(self, o): return self._bin(EX_NE, o)      # noqa: E704
    def __lt__(self, o): return self._bin(EX_LT, o)
    def __le__(self, o): return self._bin(EX_LE, o)
    def __gt__(self, o): return self._bin(EX_GT, o)
    def __ge__(self, o): return self._bin(EX_GE, o)
    def and_(self, o): return self._bin(EX_AND, o)
    def or_(self, o): return self._bin(EX_OR, o)
    def not_(self): return Expr(EX_NOT, a=self)
    __hash__ = None


def _coerce(x):
    if isinstance(x, Expr):
        return x
    if isinstance(x, bool):
        return lit(int(x))
    if isinstance(x, int):
        return lit(x)
    if isinstance(x, float):
        return litf(x)
    if x is None:
        return null()
    raise TypeError(x)


def col(i):
    return Expr(EX_COLUMN, col=i)


def lit(v):
    return Expr(EX_LIT_I64, lit_i64=v)


def litf(v):
    return Expr(EX_LIT_DOUBLE, lit_dbl=v)


def null():
    return Expr(EX_LIT_NULL)


def agg_sum(e):
    return (AGG_SUM, e)


def agg_sum1():
    return (AGG_SUM1, None)


def agg_first(e):
    """first(x): the first non-null value in scan order (udf/first.c,
    registry.cpp FirstIteration) -- arbitrary across parallel scans."""
    return (AGG_FIRST, e)


def agg_min(e):
    """min(expr) — udf/min.c"""
    from ytsaurus_amd._abi import AGG_MIN
    return (AGG_MIN, e)


def agg_max(e):
    """max(expr) — udf/max.c"""
    from ytsaurus_amd._abi import AGG_MAX
    return (AGG_MAX, e)


def agg_avg(e):
    """avg(x): double(sum)/count over non-null args
    (builtin_function_profiler.cpp avg codegen)."""
    return (AGG_AVG, e)


class Join:
    """Single equi-join spec (TJoinClause slice — see include/ytql_gpu.h
    YtJoin): foreign columns appear as columns [P .. P+len(value_cols)) to
    the rest of the plan."""

    def __init__(self, foreign_chunk, primary_key_col, foreign_key_col,
                 value_cols, is_left=False):
        self.chunk = foreign_chunk
        self.primary_key_col = primary_key_col
        self.foreign_key_col = foreign_key_col
        self.value_cols = list(value_cols)
        self.is_left = is_left

    def c_struct(self, cchunk):
        arr = (C.c_int32 * len(self.value_cols))(*self.value_cols)
        j = _abi.YtJoin(foreign=C.pointer(cchunk),
                        primary_key_col=self.primary_key_col,
                        foreign_key_col=self.foreign_key_col,
                        foreign_value_count=len(self.value_cols),
                        foreign_value_cols=arr,
                        is_left=1 if self.is_left else 0)
        j._keep = (arr, cchunk)
        return j


class Plan:
    """Mirror of the restated TQuery{WhereClause,GroupClause,OrderClause}
    shape. order_by: list of (output_column_index, desc) pairs; limit is
    required with order_by (reference QL rejects ORDER BY without LIMIT)."""

    def __init__(self, filter=None, keys=(), aggs=(), projects=(), is_merge=False,
                 order_by=(), limit=0, offset=0, with_totals=False,
                 having=None, totals_after_having=False, join=None):
        self.filter = filter
        self.keys = list(keys)
        self.aggs = list(aggs)
        self.projects = list(projects)
        self.is_merge = is_merge
        self.order_by = [(c, 1 if d else 0) for (c, d) in order_by]
        self.limit = limit
        self.offset = offset
        self.with_totals = with_totals
        self.having = having
        self.totals_after_having = totals_after_having
        self.join = join
        self._build()

    def _build(self):
        self._keep = []
        p = YtPlan()
        if self.filter is not None:
            p.filter = C.pointer(self.filter.c)
            self._keep.append(self.filter)
        p.key_count = len(self.keys)
        if self.keys:
            arr = (C.POINTER(YtExpr) * len(self.keys))()
            for i, k in enumerate(self.keys):
                arr[i] = C.pointer(k.c)
                self._keep.append(k)
            p.keys = arr
            self._keep.append(arr)
        p.agg_count = len(self.aggs)
        if self.aggs:
            aggobjs = []
            arr = (C.POINTER(YtAgg) * len(self.aggs))()
            for i, (func, arg) in enumerate(self.aggs):
                a = YtAgg(func=func)
                if arg is not None:
                    a.arg = C.pointer(arg.c)
                    self._keep.append(arg)
                aggobjs.append(a)
                arr[i] = C.pointer(a)
            p.aggs = arr
            self._keep += [aggobjs, arr]
        p.project_count = len(self.projects)
        if self.projects:
            arr = (C.POINTER(YtExpr) * len(self.projects))()
            for i, e in enumerate(self.projects):
                arr[i] = C.pointer(e.c)
                self._keep.append(e)
            p.projects = arr
            self._keep.append(arr)
        p.is_merge = 1 if self.is_merge else 0
        p.order_count = len(self.order_by)
        if self.order_by:
            oc = (C.c_int32 * len(self.order_by))(*[c for c, _ in self.order_by])
            od = (C.c_int32 * len(self.order_by))(*[d for _, d in self.order_by])
            p.order_cols = oc
            p.order_desc = od
            self._keep += [oc, od]
        p.order_limit = self.limit
        p.order_offset = self.offset
        p.with_totals = 1 if self.with_totals else 0
        p.totals_mode = 2 if self.totals_after_having else 1
        if self.having is not None:
            p.having = C.pointer(self.having.c)
            self._keep.append(self.having)
        self.c = p


# ---------------- chunk encoding ----------------

class EncodedColumn:
    """Owns a YtEncodedColumn (host blob) produced by the product encoder."""

    def __init__(self, value_type, cenc):
        self.value_type = value_type
        self._cenc = cenc
        self.segments = [cenc.segments[i] for i in range(cenc.segment_count)]

    def __del__(self):
        try:
            _abi.gpu_lib().yt_encoded_column_free(C.byref(self._cenc))
        except Exception:
            pass

    def segment_blobs(self):
        """list of (type, row_count, min_value, bytes)"""
        out = []
        for s in self.segments:
            buf = C.string_at(s.data, s.data_size)
            out.append((s.type, s.row_count, s.min_value, buf))
        return out


def _check(rc, err):
    if rc != YT_OK:
        raise RuntimeError("ytql error %d: %s" % (rc, err.value.decode()))


def encode_int64(values, nulls=None, max_segment_values=0, unsigned=False,
                 cum_rows_base=0):
    values = np.ascontiguousarray(values, dtype=np.int64)
    n = len(values)
    nullp = None
    if nulls is not None:
        nulls = np.ascontiguousarray(nulls, dtype=np.uint8)
        assert len(nulls) == n
        nullp = nulls.ctypes.data_as(C.POINTER(C.c_uint8))
    enc = YtEncodedColumn()
    err = C.create_string_buffer(256)
    rc = _abi.gpu_lib().yt_encode_int64_column(
        values.ctypes.data_as(C.POINTER(C.c_int64)), nullp, n,
        max_segment_values, 1 if unsigned else 0, cum_rows_base,
        C.byref(enc), err, 256)
    _check(rc, err)
    vt = VT_UINT64 if unsigned else VT_INT64
    return EncodedColumn(vt, enc)


def encode_string_raw(blob, begins, lens, nulls, max_segment_values=0):
    """Encode a string column from a flat byte blob + per-row (begin, len,
    null) arrays — the fast path for synthetic-data generation (numpy builds
    the blob, no per-row Python)."""
    begins = np.ascontiguousarray(begins, dtype=np.uint64)
    lens = np.ascontiguousarray(lens, dtype=np.uint32)
    nulls = np.ascontiguousarray(nulls, dtype=np.uint8)
    enc = YtEncodedColumn()
    err = C.create_string_buffer(256)
    rc = _abi.gpu_lib().yt_encode_string_column(
        blob, begins.ctypes.data_as(C.POINTER(C.c_uint64)),
        lens.ctypes.data_as(C.POINTER(C.c_uint32)),
        nulls.ctypes.data_as(C.POINTER(C.c_uint8)),
        len(begins), max_segment_values, C.byref(enc), err, 256)
    _check(rc, err)
    return EncodedColumn(VT_STRING, enc)


def encode_string(strings, max_segment_values=0):
    """strings: list of bytes/str or None (null). Encodes into the
    reference's unversioned string segment formats."""
    blobs = []
    begins = np.zeros(len(strings), dtype=np.uint64)
    lens = np.zeros(len(strings), dtype=np.uint32)
    nulls = np.zeros(len(strings), dtype=np.uint8)
    at = 0
    for i, v in enumerate(strings):
        if v is None:
            nulls[i] = 1
            continue
        if isinstance(v, str):
            v = v.encode()
        blobs.append(v)
        begins[i] = at
        lens[i] = len(v)
        at += len(v)
    blob = b"".join(blobs)
    return encode_string_raw(blob, begins, lens, nulls, max_segment_values)


class VersionedColumn:
    """Owns a YtVersionedColumn (versioned scan-format slice, SURVEY §8f
    row 3): one int64 value column + its timestamp column."""

    def __init__(self, ccol, row_count):
        self._c = ccol
        self.row_count = row_count

    def __del__(self):
        try:
            _abi.gpu_lib().yt_versioned_free(C.byref(self._c))
        except Exception:
            pass


def _encode_versioned(writes_per_row, write_ts, values, value_nulls,
                      deletes_per_row, delete_ts, max_rows_per_segment,
                      value_agg, is_double):
    wpr = np.ascontiguousarray(writes_per_row, dtype=np.uint32)
    wts = np.ascontiguousarray(write_ts, dtype=np.uint64)
    vals = np.ascontiguousarray(values,
                                dtype=np.float64 if is_double else np.int64)
    vn = np.ascontiguousarray(value_nulls, dtype=np.uint8) \
        if value_nulls is not None else None
    va = np.ascontiguousarray(value_agg, dtype=np.uint8) \
        if value_agg is not None else None
    dpr = np.ascontiguousarray(deletes_per_row, dtype=np.uint32)
    dts = np.ascontiguousarray(delete_ts, dtype=np.uint64)
    nw, nd = int(wpr.sum()), int(dpr.sum())
    if (len(wts) != nw or len(vals) != nw
            or (vn is not None and len(vn) != nw)
            or (va is not None and len(va) != nw)):
        raise ValueError("versioned encode: write_ts/values/value_nulls/"
                         f"value_agg length must equal sum(writes_per_row)={nw}")
    if len(dts) != nd:
        raise ValueError("versioned encode: delete_ts length must equal "
                         f"sum(deletes_per_row)={nd}")
    cc = _abi.YtVersionedColumn()
    err = C.create_string_buffer(256)
    fn = (_abi.gpu_lib().yt_encode_versioned_double if is_double
          else _abi.gpu_lib().yt_encode_versioned_int64)
    rc = fn(
        wpr.ctypes.data_as(C.POINTER(C.c_uint32)),
        wts.ctypes.data_as(C.POINTER(C.c_uint64)),
        vals.ctypes.data_as(C.c_void_p),
        vn.ctypes.data_as(C.POINTER(C.c_uint8)) if vn is not None else None,
        va.ctypes.data_as(C.POINTER(C.c_uint8)) if va is not None else None,
        dpr.ctypes.data_as(C.POINTER(C.c_uint32)),
        dts.ctypes.data_as(C.POINTER(C.c_uint64)),
        C.c_int64(len(wpr)), C.c_int64(max_rows_per_segment),
        C.byref(cc), err, 256)
    _check(rc, err)
    return VersionedColumn(cc, len(wpr))


def encode_versioned_int64(writes_per_row, write_ts, values, value_nulls,
                           deletes_per_row, delete_ts,
                           max_rows_per_segment=0, value_agg=None):
    """Synthetic versioned-chunk generator (byte-faithful to the reference
    timestamp/versioned-int writers: direct or dictionary values, dense or
    sparse value index, optional aggregate bitmap). write_ts DESC per row;
    values 1:1 with writes."""
    return _encode_versioned(writes_per_row, write_ts, values, value_nulls,
                             deletes_per_row, delete_ts, max_rows_per_segment,
                             value_agg, is_double=False)


def encode_versioned_double(writes_per_row, write_ts, values, value_nulls,
                            deletes_per_row, delete_ts,
                            max_rows_per_segment=0, value_agg=None):
    """Versioned DOUBLE column (floating_point_column_writer.cpp
    TVersionedFloatingPointColumnWriter layout)."""
    return _encode_versioned(writes_per_row, write_ts, values, value_nulls,
                             deletes_per_row, delete_ts, max_rows_per_segment,
                             value_agg, is_double=True)


def encode_versioned_string(writes_per_row, write_ts, values, value_nulls,
                            deletes_per_row, delete_ts,
                            max_rows_per_segment=0, value_agg=None):
    """Versioned STRING column (string_column_writer.cpp
    TVersionedStringColumnWriter; versioned Any/Composite share the byte
    layout). `values` is a list of bytes per write (b"" allowed); null
    writes (value_nulls[i] == 1) must be b""."""
    wpr = np.ascontiguousarray(writes_per_row, dtype=np.uint32)
    wts = np.ascontiguousarray(write_ts, dtype=np.uint64)
    vn = np.ascontiguousarray(value_nulls, dtype=np.uint8) \
        if value_nulls is not None else None
    va = np.ascontiguousarray(value_agg, dtype=np.uint8) \
        if value_agg is not None else None
    dpr = np.ascontiguousarray(deletes_per_row, dtype=np.uint32)
    dts = np.ascontiguousarray(delete_ts, dtype=np.uint64)
    nw, nd = int(wpr.sum()), int(dpr.sum())
    if (len(wts) != nw or len(values) != nw
            or (vn is not None and len(vn) != nw)
            or (va is not None and len(va) != nw)):
        raise ValueError("versioned string encode: write_ts/values/nulls/agg "
                         f"lengths must equal sum(writes_per_row)={nw}")
    if len(dts) != nd:
        raise ValueError("versioned string encode: delete_ts length must "
                         f"equal sum(deletes_per_row)={nd}")
    lens = np.array([len(v) for v in values], dtype=np.uint32)
    blob = b"".join(bytes(v) for v in values)
    cc = _abi.YtVersionedColumn()
    err = C.create_string_buffer(256)
    rc = _abi.gpu_lib().yt_encode_versioned_string(
        wpr.ctypes.data_as(C.POINTER(C.c_uint32)),
        wts.ctypes.data_as(C.POINTER(C.c_uint64)),
        C.c_char_p(blob),
        lens.ctypes.data_as(C.POINTER(C.c_uint32)),
        vn.ctypes.data_as(C.POINTER(C.c_uint8)) if vn is not None else None,
        va.ctypes.data_as(C.POINTER(C.c_uint8)) if va is not None else None,
        dpr.ctypes.data_as(C.POINTER(C.c_uint32)),
        dts.ctypes.data_as(C.POINTER(C.c_uint64)),
        C.c_int64(len(wpr)), C.c_int64(max_rows_per_segment),
        C.byref(cc), err, 256)
    _check(rc, err)
    return VersionedColumn(cc, len(wpr))


def _vbits_decode(vcol, bits, nulls, n):
    types = {vcol._c.val_segs[i].type for i in range(vcol._c.val_seg_count)}
    if any(t >= _abi.VSEG_STR_DIRECT_DENSE for t in types):
        # bits = (byte offset within the owning value-segment blob)<<24 | len
        segs = vcol._c.val_segs
        bounds = []
        at = 0
        for i in range(vcol._c.val_seg_count):
            at += segs[i].row_count
            bounds.append(at)
        out = []
        si = 0
        for r in range(n):
            while r >= bounds[si]:
                si += 1
            if nulls[r]:
                out.append(None)
                continue
            b = int(bits[r])
            off, ln = b >> 24, b & 0xFFFFFF
            blob = C.string_at(segs[si].data, segs[si].data_size)
            out.append(blob[off:off + ln])
        return out
    if any(t >= _abi.VSEG_DOUBLE_DENSE for t in types):
        dbl = bits.view(np.float64)
        return [None if nulls[i] else float(dbl[i]) for i in range(n)]
    return [None if nulls[i] else int(bits[i].astype(np.int64))
            for i in range(n)]


def oracle_versioned_read(vcol, timestamp, with_agg=False):
    """TEST ONLY: read the column as of `timestamp` via the oracle; returns
    (values list with None for nulls, visible bool list[, aggregate flag
    list when with_agg])."""
    n = vcol.row_count
    bits = np.zeros(max(n, 1), dtype=np.uint64)
    nulls = np.zeros(max(n, 1), dtype=np.uint8)
    vis = np.zeros(max(n, 1), dtype=np.uint8)
    agg = np.zeros(max(n, 1), dtype=np.uint8)
    err = C.create_string_buffer(256)
    rc = _abi.oracle_lib().yto_versioned_read(
        C.byref(vcol._c), C.c_uint64(timestamp),
        bits.ctypes.data_as(C.POINTER(C.c_uint64)),
        nulls.ctypes.data_as(C.POINTER(C.c_uint8)),
        vis.ctypes.data_as(C.POINTER(C.c_uint8)),
        agg.ctypes.data_as(C.POINTER(C.c_uint8)) if with_agg else None,
        err, 256)
    _check(rc, err)
    out = _vbits_decode(vcol, bits, nulls, n)
    visl = [bool(v) for v in vis[:n]]
    if with_agg:
        return out, visl, [bool(a) for a in agg[:n]]
    return out, visl


def gpu_versioned_read(vcol, timestamp, torch_mod, with_agg=False):
    """Read the versioned column as of `timestamp` on the GPU (the §8f
    row-3 slice). Returns (values list with None, visible bool list)."""
    n = vcol.row_count
    bits = torch_mod.zeros(max(n, 1), dtype=torch_mod.int64, device="cuda")
    nulls = torch_mod.zeros(max(n, 1), dtype=torch_mod.uint8, device="cuda")
    vis = torch_mod.zeros(max(n, 1), dtype=torch_mod.uint8, device="cuda")
    err = C.create_string_buffer(256)
    agg = torch_mod.zeros(max(n, 1), dtype=torch_mod.uint8, device="cuda")
    rc = _abi.gpu_lib().yt_gpu_versioned_read(
        C.byref(vcol._c), C.c_uint64(timestamp),
        C.c_void_p(bits.data_ptr()), C.c_void_p(nulls.data_ptr()),
        C.c_void_p(vis.data_ptr()),
        C.c_void_p(agg.data_ptr()) if with_agg else None,
        C.c_uint64(0), err, 256)
    _check(rc, err)
    hb = bits.cpu().numpy().astype(np.int64).view(np.uint64)
    hn = nulls.cpu().numpy()
    hv = vis.cpu().numpy()
    out = _vbits_decode(vcol, hb, hn, n)
    visl = [bool(v) for v in hv[:n]]
    if with_agg:
        ha = agg.cpu().numpy()
        return out, visl, [bool(a) for a in ha[:n]]
    return out, visl


class ScanChunk:
    """Device-resident unversioned chunk produced by the versioned-to-engine
    bridge (yt_gpu_versioned_scan_chunk); pass .chunk to gpu_execute."""

    def __init__(self, chunk, handle):
        self.chunk = chunk
        self._handle = handle
        self.row_count = chunk.row_count

    def __del__(self):
        try:
            _abi.gpu_lib().yt_gpu_scan_chunk_free(C.byref(self.chunk),
                                                  self._handle)
        except Exception:
            pass


def gpu_versioned_scan_chunk(vcol, timestamp):
    """Read `vcol` as of `timestamp` on the GPU and compact the visible rows
    into an engine-scannable device chunk (SURVEY 8f row-3 bridge)."""
    ch = _abi.YtChunk()
    handle = C.c_void_p()
    err = C.create_string_buffer(256)
    rc = _abi.gpu_lib().yt_gpu_versioned_scan_chunk(
        C.byref(vcol._c), C.c_uint64(timestamp), C.byref(ch),
        C.byref(handle), C.c_uint64(0), err, 256)
    _check(rc, err)
    return ScanChunk(ch, handle)


def gpu_versioned_scan_table(vcols, timestamp, key_chunk=None):
    """Versioned TABLE bridge: versioned value columns (+ optional
    unversioned int64 key chunk, device-encoded) read at `timestamp`,
    compacted into one engine-scannable device chunk [keys..., values...]
    (yt_gpu_versioned_scan_table)."""
    arr = (C.POINTER(_abi.YtVersionedColumn) * len(vcols))()
    for i, vc in enumerate(vcols):
        arr[i] = C.pointer(vc._c)
    ch = _abi.YtChunk()
    handle = C.c_void_p()
    err = C.create_string_buffer(256)
    rc = _abi.gpu_lib().yt_gpu_versioned_scan_table(
        arr, C.c_int32(len(vcols)),
        C.byref(key_chunk) if key_chunk is not None else None,
        C.c_uint64(timestamp), C.byref(ch), C.byref(handle),
        C.c_uint64(0), err, 256)
    _check(rc, err)
    sc = ScanChunk(ch, handle)
    sc._keep = (arr, key_chunk)
    return sc


def oracle_decode_strings(enc, n):
    """TEST ONLY: decode a string column via the oracle; returns list of
    bytes-or-None."""
    colc = YtColumn(value_type=enc.value_type,
                    segment_count=enc._cenc.segment_count,
                    segments=enc._cenc.segments)
    cap = enc._cenc.blob_size + 16
    # dictionary/RLE segments decode to more bytes than they store
    cap = max(cap, 64 * 1024 * 1024)
    blob = C.create_string_buffer(cap)
    ends = (C.c_int64 * (n + 1))()
    nulls = (C.c_uint8 * max(n, 1))()
    rc = _abi.oracle_lib().yto_decode_string_column(
        C.byref(colc), n, blob, cap, ends, nulls)
    if rc != YT_OK:
        raise RuntimeError("string decode failed rc=%d" % rc)
    raw = blob.raw[:ends[n] if n else 0]
    out = []
    for i in range(n):
        if nulls[i]:
            out.append(None)
        else:
            out.append(raw[ends[i]:ends[i + 1]])
    return out


def encode_bool(values, nulls=None, max_segment_values=0):
    values = np.ascontiguousarray(values, dtype=np.uint8)
    n = len(values)
    nullp = None
    if nulls is not None:
        nulls = np.ascontiguousarray(nulls, dtype=np.uint8)
        nullp = nulls.ctypes.data_as(C.POINTER(C.c_uint8))
    enc = YtEncodedColumn()
    err = C.create_string_buffer(256)
    rc = _abi.gpu_lib().yt_encode_bool_column(
        values.ctypes.data_as(C.POINTER(C.c_uint8)), nullp, n,
        max_segment_values, C.byref(enc), err, 256)
    _check(rc, err)
    return EncodedColumn(VT_BOOLEAN, enc)


def encode_double(values, nulls=None, max_segment_values=0):
    values = np.ascontiguousarray(values, dtype=np.float64)
    n = len(values)
    nullp = None
    if nulls is not None:
        nulls = np.ascontiguousarray(nulls, dtype=np.uint8)
        nullp = nulls.ctypes.data_as(C.POINTER(C.c_uint8))
    enc = YtEncodedColumn()
    err = C.create_string_buffer(256)
    rc = _abi.gpu_lib().yt_encode_double_column(
        values.ctypes.data_as(C.POINTER(C.c_double)), nullp, n,
        max_segment_values, C.byref(enc), err, 256)
    _check(rc, err)
    return EncodedColumn(VT_DOUBLE, enc)


class Chunk:
    """A set of encoded columns; can present host- or device-pointer views."""

    def __init__(self, columns, row_count):
        self.columns = columns       # list of EncodedColumn
        self.row_count = row_count
        self._keep = []

    def c_host(self):
        """YtChunk with host segment pointers (for the oracle)."""
        cols = (YtColumn * len(self.columns))()
        keep = [cols]
        for i, ec in enumerate(self.columns):
            nseg = ec._cenc.segment_count
            cols[i] = YtColumn(value_type=ec.value_type, segment_count=nseg,
                               segments=ec._cenc.segments)
        ch = YtChunk(row_count=self.row_count, column_count=len(self.columns),
                     columns=cols)
        self._keep.append(keep)
        return ch

    def c_device(self, torch):
        """YtChunk whose segment data pointers live in HBM. Each column's
        whole encoder blob is uploaded once as a torch uint8 tensor (torch =
        device allocator only); segment pointers are offsets into it."""
        cols = (YtColumn * len(self.columns))()
        keep = [cols]
        for i, ec in enumerate(self.columns):
            nseg = ec._cenc.segment_count
            segs = (YtSegment * nseg)()
            keep.append(segs)
            blob_base = ec._cenc.blob
            blob_size = ec._cenc.blob_size
            host = np.frombuffer(C.string_at(blob_base, blob_size), dtype=np.uint8)
            t = torch.from_numpy(host.copy()).cuda()
            keep.append(t)
            for j in range(nseg):
                s = ec._cenc.segments[j]
                off = s.data - blob_base
                segs[j] = YtSegment(type=s.type, row_count=s.row_count,
                                    min_value=s.min_value,
                                    data=t.data_ptr() + off, data_size=s.data_size)
            cols[i] = YtColumn(value_type=ec.value_type, segment_count=nseg,
                               segments=segs)
        ch = YtChunk(row_count=self.row_count, column_count=len(self.columns),
                     columns=cols)
        self._keep.append(keep)
        return ch


# ---------------- row extraction ----------------

def rows_from_rowset(rs):
    out = []
    ncols = rs.column_count
    for r in range(rs.row_count):
        row = []
        for c in range(ncols):
            v = rs.values[r * ncols + c]
            if v.type == VT_NULL:
                row.append(None)
            elif v.type == VT_INT64:
                row.append(int(C.c_int64(v.data.i64).value))
            elif v.type == VT_UINT64:
                row.append(int(v.data.u64))
            elif v.type == VT_DOUBLE:
                row.append(float(v.data.dbl))
            elif v.type == VT_BOOLEAN:
                row.append(bool(v.data.bits))
            elif v.type == VT_STRING:
                row.append(C.string_at(v.data.str, v.length))
            else:
                row.append(("?", v.type, v.data.bits))
        out.append(tuple(row))
    return out


def sort_rows(rows):
    """Order-insensitive compare helper — mirrors the reference's
    OrderedResultMatcher (unittests/evaluate/test_evaluate.cpp:174-199):
    group-by output order is hash-insert order, so both sides are sorted."""
    def key(row):
        return tuple((x is None, isinstance(x, bool), x if x is not None else 0)
                     for x in row)
    return sorted(rows, key=key)


def _mk_rowset(capacity, ncols_max=8, pool_bytes=0):
    buf = (YtValue * (capacity * ncols_max))()
    rs = YtRowset(values=buf, capacity_rows=capacity)
    rs._buf = buf  # keep alive
    if pool_bytes:
        pool = C.create_string_buffer(pool_bytes)
        rs.string_pool = C.cast(pool, C.c_char_p)
        rs.string_pool_capacity = pool_bytes
        rs._pool = pool
    return rs


# ---------------- execution ----------------

def _join_items(plan):
    j = getattr(plan, "join", None)
    if j is None:
        return []
    return list(j) if isinstance(j, (list, tuple)) else [j]


def _attach_join(plan, cchunk_factory):
    """Point plan.c.join at the YtJoin chain built over the given chunk
    flavor; returns the keep-alive object (None when the plan has no
    join). plan.join may be a Join or a list of Joins (2 items max —
    snowflake chains, later items may key on earlier items\' columns)."""
    items = _join_items(plan)
    if not items:
        plan.c.join = None
        return None
    structs = [it.c_struct(cchunk_factory(it.chunk)) for it in items]
    for i in range(len(structs) - 1):
        structs[i].next = C.pointer(structs[i + 1])
    for s in structs[1:]:
        structs[0]._keep = (structs[0]._keep, s)
    plan.c.join = C.cast(C.pointer(structs[0]), C.c_void_p)
    return structs


def _attach_join_dev(plan, join_foreign):
    """GPU flavor: join_foreign is the device YtChunk (or list of them,
    matching the plan\'s join list)."""
    items = _join_items(plan)
    if not items:
        plan.c.join = None
        return None
    assert join_foreign is not None, "plan has a join: pass join_foreign="
    devs = join_foreign if isinstance(join_foreign, (list, tuple)) \
        else [join_foreign]
    assert len(devs) == len(items), "one device chunk per join item"
    structs = [it.c_struct(devs[i]) for i, it in enumerate(items)]
    for i in range(len(structs) - 1):
        structs[i].next = C.pointer(structs[i + 1])
    for s in structs[1:]:
        structs[0]._keep = (structs[0]._keep, s)
    plan.c.join = C.cast(C.pointer(structs[0]), C.c_void_p)
    return structs


def oracle_execute(plan, chunk, nthreads=1, expect_error=False):
    """TEST/BASELINE ONLY — runs the CPU oracle restatement."""
    _j = _attach_join(plan, lambda c: c.c_host())
    ch = chunk.c_host()
    rs = _mk_rowset(max(chunk.row_count + 16, 1 << 16), pool_bytes=32 << 20)
    st = YtStatistics()
    err = C.create_string_buffer(256)
    rc = _abi.oracle_lib().yto_execute(C.byref(plan.c), C.byref(ch), C.byref(rs),
                                       C.byref(st), nthreads, err, 256)
    if expect_error:
        return rc, rows_from_rowset(rs), st
    _check(rc, err)
    return rows_from_rowset(rs), st


def oracle_partial(plan, chunk, nparts, nthreads=1):
    _j = _attach_join(plan, lambda c: c.c_host())
    ch = chunk.c_host()
    cap = chunk.row_count + 16
    states = (YtStateRow * cap)()
    counts = (C.c_int64 * nparts)()
    err = C.create_string_buffer(256)
    rc = _abi.oracle_lib().yto_partial(C.byref(plan.c), C.byref(ch), nparts,
                                       states, cap, counts, nthreads, err, 256)
    _check(rc, err)
    return states, [counts[i] for i in range(nparts)]


def oracle_partial_mk(plan, chunk, nparts, key_ranges, nthreads=1):
    """multi-key bottom query: key_ranges = (zzmin list, zzmax list) reduced
    across ranks (yt_gpu_key_ranges / oracle equivalents)."""
    _j = _attach_join(plan, lambda c: c.c_host())
    ch = chunk.c_host()
    cap = chunk.row_count + 16
    states = (YtStateRow * cap)()
    counts = (C.c_int64 * nparts)()
    kzmin = (C.c_uint64 * len(plan.keys))(*[int(v) for v in key_ranges[0]])
    kzmax = (C.c_uint64 * len(plan.keys))(*[int(v) for v in key_ranges[1]])
    err = C.create_string_buffer(256)
    rc = _abi.oracle_lib().yto_partial_mk(C.byref(plan.c), C.byref(ch), nparts,
                                          kzmin, kzmax,
                                          states, cap, counts, nthreads, err, 256)
    _check(rc, err)
    return states, [counts[i] for i in range(nparts)]


def oracle_merge_mk(plan, states_list, key_ranges, col_types=None):
    total = sum(n for _, n in states_list)
    arr = (YtStateRow * max(total, 1))()
    at = 0
    for st, n in states_list:
        for i in range(n):
            arr[at] = st[i]
            at += 1
    kzmin = (C.c_uint64 * len(plan.keys))(*[int(v) for v in key_ranges[0]])
    kzmax = (C.c_uint64 * len(plan.keys))(*[int(v) for v in key_ranges[1]])
    ct = None
    if col_types is not None:
        lst = list(col_types)[:8] + [VT_INT64] * max(0, 8 - len(col_types))
        ct = (C.c_uint8 * 8)(*lst)
    rs = _mk_rowset(max(total + 1024, 1 << 14))
    err = C.create_string_buffer(256)
    rc = _abi.oracle_lib().yto_merge_mk(C.byref(plan.c), arr, C.c_int64(total),
                                        ct, kzmin, kzmax,
                                        C.byref(rs), err, 256)
    _check(rc, err)
    return rows_from_rowset(rs)


def gpu_key_ranges(plan, device_chunk):
    k = len(plan.keys)
    kzmin = (C.c_uint64 * k)()
    kzmax = (C.c_uint64 * k)()
    err = C.create_string_buffer(256)
    rc = _abi.gpu_lib().yt_gpu_key_ranges(C.byref(plan.c), C.byref(device_chunk),
                                          kzmin, kzmax, C.c_uint64(0), err, 256)
    _check(rc, err)
    return [int(v) for v in kzmin], [int(v) for v in kzmax]


def gpu_partial_mk(plan, device_chunk, nparts, states_dev_ptr, capacity_rows,
                   key_ranges, max_groups_hint=0, stream=0):
    opts = YtExecOptions(max_groups_hint=max_groups_hint, stream=stream)
    counts = (C.c_int64 * nparts)()
    st = YtStatistics()
    k = len(plan.keys)
    kzmin = (C.c_uint64 * k)(*[int(v) for v in key_ranges[0]])
    kzmax = (C.c_uint64 * k)(*[int(v) for v in key_ranges[1]])
    err = C.create_string_buffer(512)
    rc = _abi.gpu_lib().yt_gpu_query_partial_mk(
        C.byref(plan.c), C.byref(device_chunk), C.byref(opts),
        C.c_int32(nparts), C.c_void_p(states_dev_ptr),
        C.c_int64(capacity_rows), counts, kzmin, kzmax,
        C.byref(st), err, 512)
    _check(rc, err)
    return [counts[i] for i in range(nparts)], st


def gpu_merge_mk(plan, states_dev_ptr, n_states, key_ranges, col_types=None,
                 max_groups_hint=0, stream=0, out_capacity=None):
    opts = YtExecOptions(max_groups_hint=max_groups_hint, stream=stream)
    cap = out_capacity or max(n_states + 1024, 1 << 16)
    rs = _mk_rowset(cap)
    st = YtStatistics()
    k = len(plan.keys)
    kzmin = (C.c_uint64 * k)(*[int(v) for v in key_ranges[0]])
    kzmax = (C.c_uint64 * k)(*[int(v) for v in key_ranges[1]])
    ct = None
    if col_types is not None:
        lst = list(col_types)[:8] + [VT_INT64] * max(0, 8 - len(col_types))
        ct = (C.c_uint8 * 8)(*lst)
    err = C.create_string_buffer(512)
    rc = _abi.gpu_lib().yt_gpu_merge_states_mk(
        C.byref(plan.c), C.c_void_p(states_dev_ptr), C.c_int64(n_states),
        ct, kzmin, kzmax, C.byref(opts), C.byref(rs), C.byref(st), err, 512)
    _check(rc, err)
    return rows_from_rowset(rs), st


def oracle_merge(plan, states_list):
    """states_list: list of (YtStateRow array-like, count) or a flat numpy
    structured buffer."""
    total = sum(n for _, n in states_list)
    flat = (YtStateRow * max(total, 1))()
    at = 0
    for arr, n in states_list:
        for i in range(n):
            flat[at] = arr[i]
            at += 1
    rs = _mk_rowset(max(total + 16, 1024))
    err = C.create_string_buffer(256)
    rc = _abi.oracle_lib().yto_merge(C.byref(plan.c), flat, total, C.byref(rs),
                                     err, 256)
    _check(rc, err)
    return rows_from_rowset(rs)


def coordinate_results(plan, partition_results):
    """COORDINATOR combine for the key-partitioned front queries (the
    reference's DoCoordinateAndExecute tail, ytlib/query_client/
    executor.cpp:761: each partition's front query produced its slice of
    the grouped output; the coordinator concatenates the key-disjoint
    rows, re-folds the per-partition TOTALS rows into one, and applies the
    global ORDER BY ... LIMIT over the per-partition-ordered unions —
    valid because every partition already kept its own top
    (offset+limit)). partition_results: list of row lists as returned by
    oracle_merge / gpu_merge (totals row LAST when plan.with_totals)."""
    from ytsaurus_amd._abi import (AGG_SUM, AGG_SUM1, AGG_MIN, AGG_MAX,
                                   AGG_FIRST)
    kc = len(plan.keys)
    aggs = [a if isinstance(a, tuple) else a for a in plan.aggs]
    rows = []
    totals_parts = []
    for pr in partition_results:
        pr = list(pr)
        if plan.with_totals and pr:
            totals_parts.append(pr[-1])
            pr = pr[:-1]
        rows += pr

    if plan.order_by:
        def cmp_key(r):
            out = []
            for col, desc in plan.order_by:
                v = r[col]
                null = v is None
                k = (0 if null else 1, v if not null else 0)
                if desc:
                    out.append((-k[0], _neg(v)))
                else:
                    out.append(k)
            return tuple(out)
        def _neg(v):
            if v is None:
                return 0
            if isinstance(v, bytes):
                return tuple(-b for b in v) + (1,)   # inverted memcmp
            return -v
        rows.sort(key=cmp_key)
        off = plan.offset or 0
        rows = rows[off:off + plan.limit] if plan.limit else rows[off:]

    if plan.with_totals and totals_parts:
        tot = list(totals_parts[0])
        for tp in totals_parts[1:]:
            for ai, agg in enumerate(plan.aggs):
                f = agg[0]
                i = kc + ai
                a, b = tot[i], tp[i]
                if f == AGG_SUM1:
                    tot[i] = (a or 0) + (b or 0)
                elif f == AGG_SUM:
                    if b is not None:
                        tot[i] = b if a is None else (
                            a + b if isinstance(a, float) else _wrap(a + b))
                elif f == AGG_MIN:
                    if b is not None:
                        tot[i] = b if a is None else min(a, b)
                elif f == AGG_MAX:
                    if b is not None:
                        tot[i] = b if a is None else max(a, b)
                elif f == AGG_FIRST:
                    if a is None:
                        tot[i] = b
        rows.append(tuple(tot))
    return rows


def _wrap(v):
    v &= (1 << 64) - 1
    return v - (1 << 64) if v >= (1 << 63) else v


def gpu_available():
    err = C.create_string_buffer(256)
    return _abi.gpu_lib().yt_gpu_available(err, 256) == YT_OK


def gpu_execute(plan, device_chunk, max_groups_hint=0, group_row_limit=0,
                stream=0, out_capacity=None, rowset=None, raw_rowset=False,
                join_foreign=None):
    """device_chunk: YtChunk with device pointers (Chunk.c_device).
    Pass a preallocated `rowset` (from make_rowset) to avoid per-call
    allocation; raw_rowset=True skips the Python row conversion.
    join_foreign: device YtChunk for plan.join's foreign rowset (required
    when the plan has a join)."""
    _j = _attach_join_dev(plan, join_foreign)
    opts = YtExecOptions(max_groups_hint=max_groups_hint,
                         group_row_limit=group_row_limit, stream=stream)
    if rowset is not None:
        rs = rowset
    else:
        cap = out_capacity or max(int(max_groups_hint) * 2 + 1024, 1 << 16)
        rs = _mk_rowset(cap, pool_bytes=32 << 20)
    st = YtStatistics()
    err = C.create_string_buffer(512)
    rc = _abi.gpu_lib().yt_gpu_query_execute(
        C.byref(plan.c), C.byref(device_chunk), C.byref(opts), C.byref(rs),
        C.byref(st), err, 512)
    _check(rc, err)
    if raw_rowset:
        return rs, st
    return rows_from_rowset(rs), st


def make_rowset(capacity, ncols, pool_bytes=0):
    return _mk_rowset(capacity, ncols, pool_bytes=pool_bytes)


def gpu_partial(plan, device_chunk, nparts, states_dev_ptr, capacity_rows,
                max_groups_hint=0, stream=0, join_foreign=None):
    _j = _attach_join_dev(plan, join_foreign)
    opts = YtExecOptions(max_groups_hint=max_groups_hint, stream=stream)
    counts = (C.c_int64 * nparts)()
    st = YtStatistics()
    err = C.create_string_buffer(512)
    rc = _abi.gpu_lib().yt_gpu_query_partial(
        C.byref(plan.c), C.byref(device_chunk), C.byref(opts), nparts,
        C.c_void_p(states_dev_ptr), capacity_rows, counts, C.byref(st), err, 512)
    _check(rc, err)
    return [counts[i] for i in range(nparts)], st


def oracle_partial_str(plan, chunk, nparts, nthreads=1):
    """string-keyed bottom query: returns (states ctypes array, counts list,
    pool bytes, pool_bytes-per-partition list). Each partition's states are
    contiguous; key_bits reference that partition's pool slice."""
    ch = chunk.c_host()
    cap = chunk.row_count + 16
    pool_cap = 64 + 32 * chunk.row_count
    states = (YtStateRow * cap)()
    pool = C.create_string_buffer(pool_cap)
    counts = (C.c_int64 * nparts)()
    pbytes = (C.c_int64 * nparts)()
    err = C.create_string_buffer(256)
    rc = _abi.oracle_lib().yto_partial_str(
        C.byref(plan.c), C.byref(ch), nparts, states, cap,
        C.cast(pool, C.c_char_p), pool_cap, counts, pbytes, nthreads,
        err, 256)
    _check(rc, err)
    return (states, [counts[i] for i in range(nparts)],
            pool.raw, [pbytes[i] for i in range(nparts)])


def oracle_merge_str(plan, segments, out_capacity=None, pool_capacity=None):
    """segments: list of (states ctypes array/list, count, pool bytes obj,
    pool_byte_count) received by this partition, in a fixed order."""
    nseg = len(segments)
    total = sum(s[1] for s in segments)
    allst = (YtStateRow * max(total, 1))()
    at = 0
    pool = b"".join(bytes(s[2][:s[3]]) for s in segments)
    for st_arr, cnt, _, _ in segments:
        for i in range(cnt):
            allst[at] = st_arr[i]
            at += 1
    seg_counts = (C.c_int64 * nseg)(*[s[1] for s in segments])
    seg_bytes = (C.c_int64 * nseg)(*[s[3] for s in segments])
    cap = out_capacity or (total + 16)
    pc = pool_capacity or max(len(pool), 1024)
    rs = _mk_rowset(cap, pool_bytes=pc)
    err = C.create_string_buffer(256)
    rc = _abi.oracle_lib().yto_merge_str(
        C.byref(plan.c), allst, seg_counts, nseg,
        C.c_char_p(pool), seg_bytes, C.byref(rs), err, 256)
    _check(rc, err)
    return rows_from_rowset(rs)


def gpu_partial_str(plan, device_chunk, nparts, states_dev_ptr, capacity_rows,
                    pool_dev_ptr, pool_capacity, max_groups_hint=0, stream=0):
    opts = YtExecOptions(max_groups_hint=max_groups_hint, stream=stream)
    counts = (C.c_int64 * nparts)()
    pbytes = (C.c_int64 * nparts)()
    st = YtStatistics()
    err = C.create_string_buffer(512)
    rc = _abi.gpu_lib().yt_gpu_query_partial_str(
        C.byref(plan.c), C.byref(device_chunk), C.byref(opts), nparts,
        C.c_void_p(states_dev_ptr), capacity_rows,
        C.c_void_p(pool_dev_ptr), pool_capacity, counts, pbytes,
        C.byref(st), err, 512)
    _check(rc, err)
    return ([counts[i] for i in range(nparts)],
            [pbytes[i] for i in range(nparts)], st)


def gpu_merge_str(plan, states_dev_ptr, seg_counts, pool_dev_ptr,
                  seg_pool_bytes, col_types=None, max_groups_hint=0,
                  stream=0, out_capacity=None, pool_capacity=None,
                  rowset=None, raw_rowset=False):
    opts = YtExecOptions(max_groups_hint=max_groups_hint, stream=stream)
    nseg = len(seg_counts)
    sc = (C.c_int64 * nseg)(*seg_counts)
    sb = (C.c_int64 * nseg)(*seg_pool_bytes)
    ct = None
    if col_types is not None:
        ct = (C.c_uint8 * 8)(*([int(x) for x in col_types] + [0] * (8 - len(col_types))))
    total = sum(seg_counts)
    if rowset is not None:
        rs = rowset
    else:
        cap = out_capacity or (total + 16)
        pc = pool_capacity or max(sum(seg_pool_bytes), 1024)
        rs = _mk_rowset(cap, pool_bytes=pc)
    st = YtStatistics()
    err = C.create_string_buffer(512)
    rc = _abi.gpu_lib().yt_gpu_merge_states_str(
        C.byref(plan.c), C.c_void_p(states_dev_ptr), sc, nseg,
        C.c_void_p(pool_dev_ptr), sb, ct, C.byref(opts), C.byref(rs),
        C.byref(st), err, 512)
    _check(rc, err)
    if raw_rowset:
        return rs, st
    return rows_from_rowset(rs), st


def gpu_merge(plan, states_dev_ptr, n_states, max_groups_hint=0, stream=0,
              out_capacity=None, rowset=None, raw_rowset=False,
              col_types=None):
    opts = YtExecOptions(max_groups_hint=max_groups_hint, stream=stream)
    if rowset is not None:
        rs = rowset
    else:
        cap = out_capacity or max(n_states + 1024, 1 << 16)
        rs = _mk_rowset(cap)
    st = YtStatistics()
    err = C.create_string_buffer(512)
    ct = None
    if col_types is not None:
        arr = list(col_types)[:8] + [VT_INT64] * max(0, 8 - len(col_types))
        ct = (C.c_uint8 * 8)(*arr)
    rc = _abi.gpu_lib().yt_gpu_merge_states(
        C.byref(plan.c), C.c_void_p(states_dev_ptr), n_states, ct,
        C.byref(opts), C.byref(rs), C.byref(st), err, 512)
    _check(rc, err)
    if raw_rowset:
        return rs, st
    return rows_from_rowset(rs), st
