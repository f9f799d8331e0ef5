"""ctypes mirror of include/ytql_gpu.h — the C-ABI drop-in boundary.

The Python layer is plumbing for tests/bench: all compute lives in
libytql_gpu.so (HIP, gfx950) and — for parity checking only — in
oracle/libytql_oracle.so.
"""
import ctypes as C
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_REPO = os.path.dirname(_HERE)

# status codes
YT_OK = 0
YT_ERR_INVALID_PLAN = 1
YT_ERR_INVALID_CHUNK = 2
YT_ERR_CAPACITY = 3
YT_ERR_NO_GPU = 4
YT_ERR_HIP = 5
YT_ERR_LIMIT = 6
YT_ERR_UNSUPPORTED = 7
YT_ERR_DIV_ZERO = 8

# value types
VT_NULL = 0x02
VT_INT64 = 0x03
VT_UINT64 = 0x04
VT_DOUBLE = 0x05
VT_BOOLEAN = 0x06
VT_STRING = 0x10

# segment types
SEG_DICTIONARY_RLE = 0
SEG_DICTIONARY_DENSE = 1
SEG_DIRECT_RLE = 2
SEG_DIRECT_DENSE = 3
SEG_DOUBLE = 16

# expression ops
EX_COLUMN = 0
EX_LIT_I64 = 1
EX_LIT_NULL = 2
EX_LIT_DOUBLE = 3
EX_ADD, EX_SUB, EX_MUL, EX_DIV, EX_MOD = 10, 11, 12, 13, 14
EX_EQ, EX_NE, EX_LT, EX_LE, EX_GT, EX_GE = 20, 21, 22, 23, 24, 25
EX_AND, EX_OR, EX_NOT = 30, 31, 32

AGG_SUM = 0
AGG_SUM1 = 1
AGG_MIN = 2
AGG_MAX = 3
AGG_FIRST = 4
AGG_AVG = 5


class YtValueData(C.Union):
    _fields_ = [("i64", C.c_int64), ("u64", C.c_uint64),
                ("dbl", C.c_double), ("str", C.c_char_p), ("bits", C.c_uint64)]


class YtValue(C.Structure):
    _fields_ = [("id", C.c_uint16), ("type", C.c_uint8), ("flags", C.c_uint8),
                ("length", C.c_uint32), ("data", YtValueData)]


class YtSegment(C.Structure):
    _fields_ = [("type", C.c_int32), ("row_count", C.c_int32),
                ("min_value", C.c_uint64), ("data", C.c_void_p),
                ("data_size", C.c_int64)]


class YtColumn(C.Structure):
    _fields_ = [("value_type", C.c_int32), ("segment_count", C.c_int32),
                ("segments", C.POINTER(YtSegment))]


class YtChunk(C.Structure):
    _fields_ = [("row_count", C.c_int64), ("column_count", C.c_int32),
                ("columns", C.POINTER(YtColumn))]


class YtExpr(C.Structure):
    pass


YtExpr._fields_ = [("op", C.c_int32), ("col", C.c_int32),
                   ("lit_i64", C.c_int64), ("lit_dbl", C.c_double),
                   ("a", C.POINTER(YtExpr)), ("b", C.POINTER(YtExpr))]


class YtAgg(C.Structure):
    _fields_ = [("func", C.c_int32), ("arg", C.POINTER(YtExpr))]


class YtPlan(C.Structure):
    _fields_ = [("filter", C.POINTER(YtExpr)),
                ("key_count", C.c_int32),
                ("keys", C.POINTER(C.POINTER(YtExpr))),
                ("agg_count", C.c_int32),
                ("aggs", C.POINTER(C.POINTER(YtAgg))),
                ("project_count", C.c_int32),
                ("projects", C.POINTER(C.POINTER(YtExpr))),
                ("is_merge", C.c_int32),
                ("order_count", C.c_int32),
                ("order_cols", C.POINTER(C.c_int32)),
                ("order_desc", C.POINTER(C.c_int32)),
                ("order_limit", C.c_int64),
                ("order_offset", C.c_int64),
                ("with_totals", C.c_int32),
                ("totals_mode", C.c_int32),
                ("having", C.POINTER(YtExpr)),
                ("join", C.c_void_p)]


class YtJoin(C.Structure):
    pass


YtJoin._fields_ = [("foreign", C.POINTER(YtChunk)),
                   ("primary_key_col", C.c_int32),
                   ("foreign_key_col", C.c_int32),
                   ("foreign_value_count", C.c_int32),
                   ("foreign_value_cols", C.POINTER(C.c_int32)),
                   ("is_left", C.c_int32),
                   ("next", C.POINTER(YtJoin))]


class YtExecOptions(C.Structure):
    _fields_ = [("input_row_limit", C.c_int64), ("output_row_limit", C.c_int64),
                ("group_row_limit", C.c_int64), ("device", C.c_int32),
                ("stream", C.c_uint64), ("max_groups_hint", C.c_int64)]


class YtStatistics(C.Structure):
    _fields_ = [("rows_read", C.c_int64), ("data_weight_read", C.c_int64),
                ("rows_written", C.c_int64), ("grouped_row_count", C.c_int64),
                ("incomplete_input", C.c_int32), ("incomplete_output", C.c_int32),
                ("decode_time_ms", C.c_double), ("execute_time_ms", C.c_double),
                ("kernel_scan_ms", C.c_double), ("kernel_scan_launches", C.c_int64),
                ("kernel_other_ms", C.c_double)]


class YtRowset(C.Structure):
    _fields_ = [("values", C.POINTER(YtValue)), ("capacity_rows", C.c_int64),
                ("row_count", C.c_int64), ("column_count", C.c_int32),
                ("string_pool", C.c_char_p), ("string_pool_capacity", C.c_int64),
                ("string_pool_used", C.c_int64),
                ("totals_row", C.c_int32), ("pad_", C.c_int32)]


class YtTimestampSeg(C.Structure):
    _fields_ = [("row_count", C.c_int64), ("base_timestamp", C.c_uint64),
                ("expected_writes_per_row", C.c_uint32),
                ("expected_deletes_per_row", C.c_uint32),
                ("data", C.c_void_p), ("data_size", C.c_int64)]


# versioned value segment layouts (include/ytql_gpu.h YtVersionedSegType)
VSEG_INT_DIRECT_DENSE = 0
VSEG_INT_DICT_DENSE = 1
VSEG_INT_DIRECT_SPARSE = 2
VSEG_INT_DICT_SPARSE = 3
VSEG_DOUBLE_DENSE = 16
VSEG_DOUBLE_SPARSE = 18
VSEG_STR_DIRECT_DENSE = 32
VSEG_STR_DICT_DENSE = 33
VSEG_STR_DIRECT_SPARSE = 34
VSEG_STR_DICT_SPARSE = 35
VSEG_F_AGGREGATE = 1


class YtVersionedValueSeg(C.Structure):
    _fields_ = [("row_count", C.c_int64), ("base_value", C.c_uint64),
                ("expected_values_per_row", C.c_uint32), ("type", C.c_uint32),
                ("flags", C.c_uint32), ("pad_", C.c_uint32),
                ("data", C.c_void_p), ("data_size", C.c_int64)]


class YtVersionedColumn(C.Structure):
    _fields_ = [("ts_seg_count", C.c_int32), ("val_seg_count", C.c_int32),
                ("ts_segs", C.POINTER(YtTimestampSeg)),
                ("val_segs", C.POINTER(YtVersionedValueSeg))]


class YtStateRow(C.Structure):
    _fields_ = [("key_bits", C.c_uint64), ("meta", C.c_uint64),
                ("sum_bits", C.c_uint64), ("row_count", C.c_uint64)]


class YtEncodedColumn(C.Structure):
    _fields_ = [("segment_count", C.c_int32), ("segments", C.POINTER(YtSegment)),
                ("blob", C.c_void_p), ("blob_size", C.c_int64)]


def _sig(lib, name, res, args):
    fn = getattr(lib, name)
    fn.restype = res
    fn.argtypes = args
    return fn


_gpu_lib = None
_oracle_lib = None


def gpu_lib():
    """The product library. Raises if not built — no silent fallback."""
    global _gpu_lib
    if _gpu_lib is None:
        # bind the HIP runtime torch already loaded (torch ships its own
        # libamdhip64; loading ours first would map a second runtime that
        # sees no device once torch initializes)
        try:
            import torch  # noqa: F401
        except Exception:
            pass
        path = os.path.join(_HERE, "libytql_gpu.so")
        if not os.path.exists(path):
            raise RuntimeError(
                "libytql_gpu.so not built; run python -c 'import __graft_entry__; __graft_entry__.build()'")
        lib = C.CDLL(path)
        _sig(lib, "yt_gpu_available", C.c_int, [C.c_char_p, C.c_size_t])
        _sig(lib, "yt_gpu_pool_trim", None, [])
        _sig(lib, "yt_gpu_query_execute", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk), C.POINTER(YtExecOptions),
              C.POINTER(YtRowset), C.POINTER(YtStatistics), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_gpu_query_partial", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk), C.POINTER(YtExecOptions),
              C.c_int32, C.c_void_p, C.c_int64, C.POINTER(C.c_int64),
              C.POINTER(YtStatistics), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_gpu_merge_states", C.c_int,
             [C.POINTER(YtPlan), C.c_void_p, C.c_int64,
              C.POINTER(C.c_uint8), C.POINTER(YtExecOptions),
              C.POINTER(YtRowset), C.POINTER(YtStatistics), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_gpu_merge_states_mk", C.c_int,
             [C.POINTER(YtPlan), C.c_void_p, C.c_int64,
              C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
              C.POINTER(YtExecOptions),
              C.POINTER(YtRowset), C.POINTER(YtStatistics), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_gpu_key_ranges", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk),
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
              C.c_uint64, C.c_char_p, C.c_size_t])
        _sig(lib, "yt_gpu_query_partial_mk", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk), C.POINTER(YtExecOptions),
              C.c_int32, C.c_void_p, C.c_int64, C.POINTER(C.c_int64),
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
              C.POINTER(YtStatistics), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_gpu_query_partial_str", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk), C.POINTER(YtExecOptions),
              C.c_int32, C.c_void_p, C.c_int64, C.c_void_p, C.c_int64,
              C.POINTER(C.c_int64), C.POINTER(C.c_int64),
              C.POINTER(YtStatistics), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_gpu_merge_states_str", C.c_int,
             [C.POINTER(YtPlan), C.c_void_p, C.POINTER(C.c_int64), C.c_int32,
              C.c_void_p, C.POINTER(C.c_int64), C.POINTER(C.c_uint8),
              C.POINTER(YtExecOptions), C.POINTER(YtRowset),
              C.POINTER(YtStatistics), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_encode_int64_column", C.c_int,
             [C.POINTER(C.c_int64), C.POINTER(C.c_uint8), C.c_int64, C.c_int32,
              C.c_int32, C.c_int64, C.POINTER(YtEncodedColumn), C.c_char_p,
              C.c_size_t])
        _sig(lib, "yt_encode_double_column", C.c_int,
             [C.POINTER(C.c_double), C.POINTER(C.c_uint8), C.c_int64, C.c_int32,
              C.POINTER(YtEncodedColumn), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_encode_bool_column", C.c_int,
             [C.POINTER(C.c_uint8), C.POINTER(C.c_uint8), C.c_int64, C.c_int32,
              C.POINTER(YtEncodedColumn), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_encode_string_column", C.c_int,
             [C.c_char_p, C.POINTER(C.c_uint64), C.POINTER(C.c_uint32),
              C.POINTER(C.c_uint8), C.c_int64, C.c_int32,
              C.POINTER(YtEncodedColumn), C.c_char_p, C.c_size_t])
        _sig(lib, "yt_encoded_column_free", None, [C.POINTER(YtEncodedColumn)])
        _sig(lib, "yt_bitpack_size_words", C.c_int64, [C.c_uint64, C.c_int64])
        _sig(lib, "yt_bitpack", C.c_int64,
             [C.POINTER(C.c_uint64), C.c_int64, C.c_uint64, C.POINTER(C.c_uint64)])
        _gpu_lib = lib
    return _gpu_lib


def oracle_lib():
    """TEST INFRASTRUCTURE: the CPU parity oracle (oracle/ytql_oracle.c).
    Only tests/, smoke() and bench.py's cpu_baseline leg may call this."""
    global _oracle_lib
    if _oracle_lib is None:
        path = os.path.join(_REPO, "oracle", "libytql_oracle.so")
        if not os.path.exists(path):
            raise RuntimeError("oracle not built; run __graft_entry__.build()")
        lib = C.CDLL(path)
        _sig(lib, "yto_execute", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk), C.POINTER(YtRowset),
              C.POINTER(YtStatistics), C.c_int, C.c_char_p, C.c_size_t])
        _sig(lib, "yto_decode_column", C.c_int,
             [C.POINTER(YtColumn), C.c_int64, C.POINTER(C.c_int64), C.POINTER(C.c_uint8)])
        _sig(lib, "yto_partial", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk), C.c_int32,
              C.POINTER(YtStateRow), C.c_int64, C.POINTER(C.c_int64),
              C.c_int, C.c_char_p, C.c_size_t])
        _sig(lib, "yto_merge", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtStateRow), C.c_int64,
              C.POINTER(YtRowset), C.c_char_p, C.c_size_t])
        _sig(lib, "yto_partial_mk", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk), C.c_int32,
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
              C.POINTER(YtStateRow), C.c_int64, C.POINTER(C.c_int64),
              C.c_int, C.c_char_p, C.c_size_t])
        _sig(lib, "yto_merge_mk", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtStateRow), C.c_int64,
              C.POINTER(C.c_uint8),
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
              C.POINTER(YtRowset), C.c_char_p, C.c_size_t])
        _sig(lib, "yto_partial_str", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtChunk), C.c_int32,
              C.POINTER(YtStateRow), C.c_int64,
              C.c_char_p, C.c_int64,
              C.POINTER(C.c_int64), C.POINTER(C.c_int64),
              C.c_int, C.c_char_p, C.c_size_t])
        _sig(lib, "yto_merge_str", C.c_int,
             [C.POINTER(YtPlan), C.POINTER(YtStateRow),
              C.POINTER(C.c_int64), C.c_int,
              C.c_char_p, C.POINTER(C.c_int64),
              C.POINTER(YtRowset), C.c_char_p, C.c_size_t])
        _sig(lib, "yto_partition_hash", C.c_uint64, [C.c_uint64, C.c_int])
        _sig(lib, "yto_bitunpack", C.c_int64,
             [C.c_void_p, C.POINTER(C.c_uint64), C.c_int64])
        _sig(lib, "yto_decode_string_column", C.c_int,
             [C.POINTER(YtColumn), C.c_int64, C.c_char_p, C.c_int64,
              C.POINTER(C.c_int64), C.POINTER(C.c_uint8)])
        _oracle_lib = lib
    return _oracle_lib
