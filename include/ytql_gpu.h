/* ytql_gpu.h — C-ABI drop-in boundary for the YTsaurus dynamic-table query
 * engine hot path (scan → filter → hash-aggregate) on AMD MI355X (gfx950).
 *
 * This header declares the seam that replaces, for this path:
 *   - IEvaluator::Run                (reference: yt/yt/library/query/engine_api/evaluator.h;
 *                                     impl yt/yt/library/query/engine/evaluator.cpp:51-105)
 *   - TCGQuerySignature              (engine_api/evaluation_helpers.h:330)
 *   - TExecutionContext              (engine_api/evaluation_helpers.h:251-280)
 *   - TQueryStatistics               (yt/yt/client/query_client/query_statistics.h:49-79)
 *   - the columnar-batch ingress     (yt/yt/client/table_client/row_batch.h:39-202)
 *   - the unversioned column writer  (yt/yt/ytlib/table_chunk_format/integer_column_writer.cpp)
 *     (synthetic-chunk generator; encoded bytes match the reference formats)
 *
 * Plain pointers and sizes only; no C++ or torch types cross this boundary.
 * Errors: non-zero status + message copied into the caller's buffer
 * (the reference uses C++ exceptions; TInterruptedIncompleteException maps to
 * statistics.incomplete_output = 1 with status YT_OK, mirroring
 * engine/cg_routines/registry.cpp:1902-1907).
 */
#ifndef YTQL_GPU_H
#define YTQL_GPU_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status codes ---- */
enum {
    YT_OK = 0,
    YT_ERR_INVALID_PLAN = 1,
    YT_ERR_INVALID_CHUNK = 2,
    YT_ERR_CAPACITY = 3,      /* caller-provided output buffer too small */
    YT_ERR_NO_GPU = 4,        /* HIP device unavailable — NO CPU fallback exists */
    YT_ERR_HIP = 5,           /* HIP runtime error (message has details) */
    YT_ERR_LIMIT = 6,         /* hard limit exceeded (non-interrupt kind) */
    YT_ERR_UNSUPPORTED = 7,
    YT_ERR_DIV_ZERO = 8,      /* integer division/modulo by zero in an expression */
};

/* ---- value model ----
 * 16-byte mirror of TUnversionedValue (client/table_client/unversioned_value.h)
 * / TPIValue (engine_api/position_independent_value.h:37-66). */
enum {
    YT_VT_NULL = 0x02,    /* EValueType::Null */
    YT_VT_INT64 = 0x03,
    YT_VT_UINT64 = 0x04,
    YT_VT_DOUBLE = 0x05,
    YT_VT_BOOLEAN = 0x06,
    YT_VT_STRING = 0x10,
};

typedef struct YtValue {
    uint16_t id;
    uint8_t type;     /* YT_VT_* */
    uint8_t flags;
    uint32_t length;  /* string length */
    union {
        int64_t i64;
        uint64_t u64;
        double dbl;
        const char* str;
        uint64_t bits;
    } data;
} YtValue;

/* ---- encoded column segments ----
 * Byte layouts identical to the reference's unversioned segment formats
 * (table_chunk_format/integer_column_writer.cpp, private.h:25-30,
 *  floating_point_column_writer.cpp). */
enum {
    YT_SEG_DICTIONARY_RLE = 0,
    YT_SEG_DICTIONARY_DENSE = 1,
    YT_SEG_DIRECT_RLE = 2,
    YT_SEG_DIRECT_DENSE = 3,
    YT_SEG_DOUBLE = 16,       /* unversioned double segment (single layout) */
    YT_SEG_BOOLEAN = 17,      /* unversioned boolean segment (single layout) */
};

typedef struct YtSegment {
    int32_t type;         /* YT_SEG_* */
    int32_t row_count;    /* rows covered by this segment (<= 128*1024) */
    uint64_t min_value;   /* TIntegerSegmentMeta.min_value (zigzag space) */
    const void* data;     /* segment blob (device ptr for GPU entries, host ptr for oracle/encode) */
    int64_t data_size;    /* bytes */
} YtSegment;

typedef struct YtColumn {
    int32_t value_type;   /* YT_VT_INT64 / YT_VT_UINT64 / YT_VT_DOUBLE / YT_VT_BOOLEAN */
    int32_t segment_count;
    const YtSegment* segments;
} YtColumn;

typedef struct YtChunk {
    int64_t row_count;
    int32_t column_count;
    const YtColumn* columns;
} YtChunk;

/* ---- plan ----
 * Restates TQuery{WhereClause,GroupClause} (base/query.h:483-578) for the
 * scan→filter→group-by shape; expression grammar covers the hot-path tests.
 */
enum {
    YT_EX_COLUMN = 0,     /* leaf: input column[col] */
    YT_EX_LIT_I64 = 1,
    YT_EX_LIT_NULL = 2,
    YT_EX_LIT_DOUBLE = 3,
    YT_EX_ADD = 10, YT_EX_SUB = 11, YT_EX_MUL = 12, YT_EX_DIV = 13, YT_EX_MOD = 14,
    YT_EX_EQ = 20, YT_EX_NE = 21, YT_EX_LT = 22, YT_EX_LE = 23, YT_EX_GT = 24, YT_EX_GE = 25,
    YT_EX_AND = 30, YT_EX_OR = 31, YT_EX_NOT = 32,
};

typedef struct YtExpr {
    int32_t op;           /* YT_EX_* */
    int32_t col;          /* for YT_EX_COLUMN */
    int64_t lit_i64;      /* for YT_EX_LIT_I64 */
    double lit_dbl;       /* for YT_EX_LIT_DOUBLE */
    const struct YtExpr* a;
    const struct YtExpr* b;
} YtExpr;

enum {
    YT_AGG_SUM = 0,       /* udf/sum.c — null-propagating add; int64 wraps mod 2^64 */
    YT_AGG_SUM1 = 1,      /* sum(1) == count; QL has no count aggregate (ql_query_ut.cpp:3200) */
    YT_AGG_MIN = 2,       /* udf/min.c */
    YT_AGG_MAX = 3,       /* udf/max.c */
    YT_AGG_FIRST = 4,     /* udf/first.c + registry.cpp FirstIteration:3642-3663:
                             the first NON-NULL value in scan order; scan order
                             across GPU threads is nondeterministic (as it is
                             across the reference's tablets), so with several
                             distinct values per group the pick is arbitrary.
                             String arguments: oracle only (GPU refuses). */
    YT_AGG_AVG = 5,       /* builtin_function_profiler.cpp:1483-1607 avg:
                             state {i64 count, arg-typed sum (int wraps,
                             double fadd)}; finalize null when count==0 else
                             double(sum)/count */
};

typedef struct YtAgg {
    int32_t func;         /* YT_AGG_* */
    const YtExpr* arg;    /* NULL for YT_AGG_SUM1 */
} YtAgg;

typedef struct YtPlan {
    const YtExpr* filter;         /* NULL = no WHERE clause */
    int32_t key_count;            /* 0 + agg_count>0 = global aggregate (one group) */
    const YtExpr* const* keys;
    int32_t agg_count;
    const YtAgg* const* aggs;
    int32_t project_count;        /* used when agg_count == 0 (plain scan): projected exprs */
    const YtExpr* const* projects;
    int32_t is_merge;             /* front-query mode: input = [keys..., states...] rows
                                     (cg_fragment_compiler.cpp:4016-4124) */
    /* ORDER BY ... LIMIT — TTopCollector semantics (engine_api/top_collector
     * -inl.h AddRow: keep the (offset+limit) least rows by the lexicographic
     * comparer, emit sorted ascending from index offset; registry.cpp
     * OrderOpHelper:1948-1997). order_cols index the OUTPUT row of this plan
     * stage (projections in scan mode, [keys..., aggs...] in group mode).
     * Per-key comparison mirrors the codegen universal comparer
     * (cg_fragment_compiler.cpp:400-530): null < any value, int64 signed,
     * uint64/boolean unsigned, double by value (NaN comparison = error),
     * string memcmp with length tiebreak; a descending key inverts the
     * outcome (nulls go last). LIMIT is required with ORDER BY, as in the
     * reference QL. */
    int32_t order_count;
    const int32_t* order_cols;    /* output-column indices */
    const int32_t* order_desc;    /* 0 = ascending, 1 = descending */
    int64_t order_limit;
    int64_t order_offset;
    /* WITH TOTALS (TotalsMode::BeforeHaving; registry.cpp
     * TGroupByClosure::InsertTotals/FlushTotals:1556-1650, test semantics
     * ql_query_ut.cpp:3432-3510): one extra output row with null group keys
     * whose aggregates cover ALL grouped rows (including the null-key
     * group), computed before any ORDER BY/limit slicing. The totals row is
     * appended last and flagged in YtRowset.totals_row — it mirrors the
     * reference's separate EStreamTag::Totals stream. */
    int32_t with_totals;
    /* 1 = BeforeHaving (default), 2 = AfterHaving — syntactic position of
     * WITH TOTALS relative to HAVING (parser.ypp:469-481): BeforeHaving
     * totals cover all groups, AfterHaving totals only the groups that
     * survive the having filter (folding_profiler.cpp:1810-1815). */
    int32_t totals_mode;
    /* HAVING clause (TQuery::HavingClause, base/query.h:499): a predicate
     * over the group output row [keys..., aggs...], applied after
     * aggregation and before ORDER BY/limit. */
    const YtExpr* having;
    /* Equi-join against a foreign rowset (TJoinClause, base/query.h:364-417;
     * runtime semantics cg_routines/registry.cpp MultiJoinOpHelper:599-960:
     * joined row = [primary columns..., foreign columns...], INNER drops
     * unmatched primaries, LEFT null-extends, null keys join null keys —
     * the codegen eq-comparer treats null == null). This round: one join,
     * UNIQUE foreign keys (the dimension-lookup case; duplicate keys fail
     * loudly — the reference's cross-product expansion is round-3 work),
     * int64/uint64/boolean key columns. The foreign value columns appear to
     * the rest of the plan as columns [P .. P+foreign_value_count) where P =
     * the primary chunk's column count — filter/keys/aggs/order may
     * reference them; the join applies BEFORE the WHERE clause, as in the
     * reference pipeline. */
    const struct YtJoin* join;
} YtPlan;

typedef struct YtJoin {
    const YtChunk* foreign;          /* encoded foreign rowset */
    int32_t primary_key_col;         /* equality key column: an index into the
                                        row AS EXTENDED SO FAR — primary
                                        columns plus every earlier join item's
                                        appended values — so a later item can
                                        key on an earlier item's output
                                        (snowflake chains, the reference's
                                        TMultiJoinParameters item list,
                                        registry.cpp MultiJoinOpHelper) */
    int32_t foreign_key_col;         /* equality key column in `foreign` */
    int32_t foreign_value_count;
    const int32_t* foreign_value_cols;  /* foreign columns appended to the row */
    int32_t is_left;                 /* 0 = INNER, 1 = LEFT */
    const struct YtJoin* next;       /* next join item (2 items max this
                                        round; duplicate foreign keys only on
                                        the FIRST item) */
} YtJoin;

/* ---- execution context / statistics ----
 * Mirrors TExecutionContext limits and TQueryStatistics counters. */
typedef struct YtExecOptions {
    int64_t input_row_limit;      /* 0 = unlimited */
    int64_t output_row_limit;
    int64_t group_row_limit;
    int32_t device;               /* HIP device ordinal */
    uint64_t stream;              /* hipStream_t as integer; 0 = default stream */
    int64_t max_groups_hint;      /* sizes the device hash table; 0 = default */
} YtExecOptions;

typedef struct YtStatistics {
    int64_t rows_read;
    int64_t data_weight_read;     /* encoded bytes consumed */
    int64_t rows_written;
    int64_t grouped_row_count;
    int32_t incomplete_input;
    int32_t incomplete_output;
    double decode_time_ms;        /* host wall around device pipeline */
    double execute_time_ms;
    /* device-event timings of the dominant kernels (for roofline evidence) */
    double kernel_scan_ms;        /* total GPU time in scan_*_agg kernels */
    int64_t kernel_scan_launches;
    double kernel_other_ms;
} YtStatistics;

/* ---- result rowset ----
 * Caller provides capacity; library writes row-major 16-byte YtValues.
 * (Matches the writer seam IUnversionedRowsetWriter::Write,
 *  client/table_client/unversioned_writer.h:21-32.) */
typedef struct YtRowset {
    YtValue* values;              /* capacity_rows * column_count values, row-major */
    int64_t capacity_rows;
    int64_t row_count;            /* out */
    int32_t column_count;         /* out */
    char* string_pool;            /* optional pool for string payloads */
    int64_t string_pool_capacity;
    int64_t string_pool_used;     /* out */
    int32_t totals_row;           /* out: 1 = the LAST row is the WITH TOTALS
                                     stream row (EStreamTag::Totals) */
    int32_t pad_;
} YtRowset;

/* ---- versioned scan-format slice ----
 * The MVCC "scan format" (ytlib/columnar_chunk_format) for flat tables:
 * a timestamp segment (table_chunk_format/timestamp_writer.cpp DumpSegment:
 * blob = five header-carrying bit-packed vectors
 * [ts dictionary − base][write ts ids][delete ts ids]
 * [cumulative write counts, diff-from-expected][cumulative delete counts,
 * diff-from-expected]; meta = TTimestampMeta base/expected fields,
 * prepared_meta.h:40-60) and versioned value segments (DirectDense int64
 * this round: integer_column_writer.cpp:119-246 + column_writer_detail.cpp
 * DumpVersionedData: blob = [cumulative values-per-row diffs][per-value
 * timestamp index][values (zigzag − base)][null bitmap]; values correspond
 * 1:1 with the row's write timestamps in this slice). Timestamps within a
 * row are strictly descending (rowset_builder.cpp:1052-1060 asserts).
 * Read-at-timestamp semantics (rowset_builder.cpp:1042-1166
 * TRowAllocatorBase::DoAllocateRow, produceAll = false): latest write ≤ T
 * that is newer than the latest delete ≤ T; no such write → the row is not
 * visible at T. */
typedef struct YtTimestampSeg {
    int64_t row_count;
    uint64_t base_timestamp;
    uint32_t expected_writes_per_row;
    uint32_t expected_deletes_per_row;
    const void* data;
    int64_t data_size;
} YtTimestampSeg;

/* versioned value segment layouts (round 2 widens the slice):
 * value-index part (column_writer_detail.cpp DumpVersionedData): Dense =
 * [cumulative values-per-row, diff-from-expected] iff denseSize <=
 * sparseSize, else Sparse = [row index per value] (bit-packed, max =
 * last row index). Value part (integer/floating_point writers): int
 * Direct = [packed zigzag-base values][null bitmap]; int Dictionary =
 * [packed dictionary (first-appearance, value-min)][packed ids, 0=null]
 * (dictionarySize < directSize rule, integer_column_writer.cpp:222-239);
 * double = [u64 count][raw doubles][null bitmap]. An aggregate-column
 * bitmap (EValueFlags::Aggregate per value) sits between the timestamp
 * ids and the value part when YT_VSEG_F_AGGREGATE is set. */
enum YtVersionedSegType {
    YT_VSEG_INT_DIRECT_DENSE = 0,
    YT_VSEG_INT_DICT_DENSE = 1,
    YT_VSEG_INT_DIRECT_SPARSE = 2,
    YT_VSEG_INT_DICT_SPARSE = 3,
    YT_VSEG_DOUBLE_DENSE = 16,
    YT_VSEG_DOUBLE_SPARSE = 18,
    /* versioned STRING value segments (string_column_writer.cpp
     * TVersionedStringColumnWriter::DumpSegment:325-360; the Any and
     * Composite versioned writers share this byte layout :365-401).
     * Value part — Direct (DumpDirectValues:205-229): [packed cumulative
     * END offsets, diff-from-expected (null values step 0)][null bitmap]
     * [value bytes]; Dictionary (DumpDictionaryValues:153-203):
     * [packed ids, 0 = null, first-appearance 1-based][packed cumulative
     * dictionary END offsets, diff-from-expected][dictionary bytes].
     * Chosen by dictionaryByteSize < directByteSize (:338-352, size
     * estimates :80-92). base_value carries
     * TStringSegmentMeta.expected_length. The read entry points return
     * bits = (byte offset within the value-segment blob) << 24 | length. */
    YT_VSEG_STR_DIRECT_DENSE = 32,
    YT_VSEG_STR_DICT_DENSE = 33,
    YT_VSEG_STR_DIRECT_SPARSE = 34,
    YT_VSEG_STR_DICT_SPARSE = 35,
};
enum { YT_VSEG_F_AGGREGATE = 1 };     /* aggregate bitmap present */

typedef struct YtVersionedValueSeg {
    int64_t row_count;
    uint64_t base_value;              /* zigzag-space min over non-null values */
    uint32_t expected_values_per_row; /* dense index layouts only */
    uint32_t type;                    /* YT_VSEG_* */
    uint32_t flags;                   /* YT_VSEG_F_* */
    uint32_t pad_;
    const void* data;
    int64_t data_size;
} YtVersionedValueSeg;

typedef struct YtVersionedColumn {
    int32_t ts_seg_count;
    int32_t val_seg_count;            /* == ts_seg_count, same row split */
    const YtTimestampSeg* ts_segs;
    const YtVersionedValueSeg* val_segs;
} YtVersionedColumn;

/* Encode one versioned int64 column + its timestamp column from flattened
 * per-row write/delete lists (write_ts DESC within each row; values[i]
 * belongs to the i-th write of its row). Synthetic-data generator, mirrors
 * the reference writers byte-for-byte. Caller frees with
 * yt_versioned_free. */
int yt_encode_versioned_int64(
    const uint32_t* writes_per_row, const uint64_t* write_ts,
    const int64_t* values, const uint8_t* value_nulls,
    const uint8_t* value_agg,         /* optional per-value aggregate flags */
    const uint32_t* deletes_per_row, const uint64_t* delete_ts,
    int64_t row_count, int64_t max_rows_per_segment,
    YtVersionedColumn* out, char* errbuf, size_t errlen);
/* Same for a versioned DOUBLE column (floating_point_column_writer.cpp
 * TVersionedFloatingPointColumnWriter: value part = [u64 count]
 * [raw IEEE doubles][null bitmap]). */
int yt_encode_versioned_double(
    const uint32_t* writes_per_row, const uint64_t* write_ts,
    const double* values, const uint8_t* value_nulls,
    const uint8_t* value_agg,
    const uint32_t* deletes_per_row, const uint64_t* delete_ts,
    int64_t row_count, int64_t max_rows_per_segment,
    YtVersionedColumn* out, char* errbuf, size_t errlen);
/* Versioned STRING column (also the byte layout of versioned Any /
 * Composite columns, string_column_writer.cpp:365-401): value_bytes is the
 * concatenation of every write's bytes in flattened write order;
 * value_lens[i] is the i-th write's length (0 for null writes). */
int yt_encode_versioned_string(
    const uint32_t* writes_per_row, const uint64_t* write_ts,
    const char* value_bytes, const uint32_t* value_lens,
    const uint8_t* value_nulls, const uint8_t* value_agg,
    const uint32_t* deletes_per_row, const uint64_t* delete_ts,
    int64_t row_count, int64_t max_rows_per_segment,
    YtVersionedColumn* out, char* errbuf, size_t errlen);
void yt_versioned_free(YtVersionedColumn* col);

/* Read the column as of timestamp T on the GPU: out_visible[r] = 1 iff the
 * row has a visible write at T; out_bits/out_null = the column value of the
 * visible version. All out pointers are DEVICE memory sized row_count. */
int yt_gpu_versioned_read(
    const YtVersionedColumn* col, uint64_t timestamp,
    uint64_t* out_bits, uint8_t* out_null, uint8_t* out_visible,
    uint8_t* out_agg,                 /* optional: visible value's aggregate flag */
    uint64_t stream, char* errbuf, size_t errlen);

/* Read the column at `timestamp` and compact the VISIBLE rows into a
 * device-resident single-column unversioned chunk — reference-layout
 * DirectDense width-64 segments (128 Ki row cap, min_value = 0, values in
 * zigzag space) that yt_gpu_query_execute scans directly. This is the
 * versioned→engine bridge: deleted / not-yet-written rows disappear, row
 * order is preserved. out_chunk's column/segment arrays are heap-allocated
 * host structs pointing at device blobs; release BOTH with
 * yt_gpu_scan_chunk_free. */
int yt_gpu_versioned_scan_chunk(
    const YtVersionedColumn* col, uint64_t timestamp,
    YtChunk* out_chunk, void** out_handle,
    uint64_t stream, char* errbuf, size_t errlen);

/* Versioned TABLE bridge: nvcols versioned value columns (int64 or
 * double; all sharing the table's per-row write/delete timestamp lists)
 * plus optional unversioned int64 KEY columns (keys in the scan format
 * are plain unversioned segments — rowset_builder.cpp key readers), read
 * at `timestamp` and compacted into one device-resident unversioned
 * chunk [keys..., values...] for yt_gpu_query_execute. Free with
 * yt_gpu_scan_chunk_free. */
int yt_gpu_versioned_scan_table(
    const YtVersionedColumn* const* vcols, int nvcols,
    const YtChunk* key_chunk, uint64_t timestamp,
    YtChunk* out_chunk, void** out_handle,
    uint64_t stream, char* errbuf, size_t errlen);
void yt_gpu_scan_chunk_free(YtChunk* chunk, void* handle);

/* =========================== entry points =========================== */

/* Library/device probe. Returns YT_OK when a gfx950 HIP device is usable. */
int yt_gpu_available(char* errbuf, size_t errlen);

/* Releases the library's cached device/pinned buffers (queries reuse large
 * allocations across calls, mirroring the reference evaluator's pooling). */
void yt_gpu_pool_trim(void);

/* The evaluator seam (replaces IEvaluator::Run for this plan shape).
 * Chunk segment data pointers must be DEVICE pointers (HBM-resident);
 * output rowset buffers are HOST memory. Synchronous on `options->stream`. */
int yt_gpu_query_execute(
    const YtPlan* plan,
    const YtChunk* chunk,
    const YtExecOptions* options,
    YtRowset* output,
    YtStatistics* stats,
    char* errbuf, size_t errlen);

/* Two-phase (coordinated) path, mirroring the bottom/front query split
 * (engine_api/coordinator.h:26,78,89; shuffle engine_api/shuffling_reader.cpp:21-88).
 *
 * yt_gpu_query_partial: bottom query — scan+filter+group into partial states,
 * then hash-partition state rows by hash(group key) % partition_count into
 * caller's DEVICE buffer `states` (capacity_rows state rows). A state row is
 * (key_count + 3*agg_count... ) — packed as YtStateRow records below.
 * part_offsets/part_counts (length partition_count, host) describe the layout.
 */
/* State-row record: {group key, ONE value-carrying aggregate state, sum(1)}
 * plans — the value slot is sum(x) / avg(x) / min(x) / max(x); the plan
 * the merge receives names the function (the reference's merge codegen is
 * likewise driven by the front query's aggregate list,
 * cg_fragment_compiler.cpp:4016-4134). avg's state is exactly the
 * reference's coordinated {count,sum} (GroupByWithAvgCoordinated
 * ql_query_ut.cpp:2760). Multi-key plans pack composites into key_bits
 * (the _mk entries); string keys reference an exchanged pool slice (the
 * _str entries). 32 bytes, device-native layout. */
typedef struct YtStateRow {
    uint64_t key_bits;               /* raw key bits / packed composite / pool ref */
    uint64_t meta;                   /* bit0 = key null; bit1 = value is double;
                                        bits 8..63 = EXACT non-null arg count */
    uint64_t sum_bits;               /* running sum (wrapping i64 / double bits)
                                        or running min/max */
    uint64_t row_count;              /* rows in group == sum(1) state */
} YtStateRow;

int yt_gpu_query_partial(
    const YtPlan* plan,
    const YtChunk* chunk,
    const YtExecOptions* options,
    int32_t partition_count,
    void* states_device,             /* device buffer for partitioned YtStateRow[] */
    int64_t capacity_rows,
    int64_t* part_counts,            /* out, host, length partition_count */
    YtStatistics* stats,
    char* errbuf, size_t errlen);

/* Multi-key sharding support: a multi-key GROUP BY packs its key columns'
 * zigzag-space values into one <=62-bit composite (DevPlan kp_*, mirroring
 * the coordinated shuffle's key-prefix hash, shuffling_reader.cpp:40-42).
 * Across ranks the packing BASES must agree, so the caller
 *   (1) asks each rank for its local per-key-column zigzag ranges,
 *   (2) all-reduces min over zzmin and max over zzmax,
 *   (3) passes the common ranges to partial AND merge.
 * Single-key plans ignore the ranges. key_zzmin/key_zzmax have
 * plan->key_count entries; each key must be a plain int64/uint64/boolean
 * column (the single-GPU composite-key rule). */
int yt_gpu_key_ranges(
    const YtPlan* plan, const YtChunk* chunk,
    uint64_t* key_zzmin, uint64_t* key_zzmax,   /* out, length key_count */
    uint64_t stream, char* errbuf, size_t errlen);

int yt_gpu_query_partial_mk(
    const YtPlan* plan, const YtChunk* chunk, const YtExecOptions* options,
    int32_t partition_count, void* states_device, int64_t capacity_rows,
    int64_t* part_counts,
    const uint64_t* key_zzmin, const uint64_t* key_zzmax,  /* common ranges */
    YtStatistics* stats, char* errbuf, size_t errlen);

/* front query: merge state rows (device buffer, e.g. post all-to-all) and
 * finalize into `output` (host). Implements Merge+finalize semantics of
 * cg_fragment_compiler.cpp:4116-4134 + udf/sum.c:47-65. */
int yt_gpu_merge_states(
    const YtPlan* plan,
    const void* states_device,
    int64_t state_row_count,
    const uint8_t* col_types,        /* optional YT_VT_* per bottom-query column
                                        (double sums finalize as doubles); NULL
                                        = all int64 */
    const YtExecOptions* options,
    YtRowset* output,
    YtStatistics* stats,
    char* errbuf, size_t errlen);

/* multi-key front merge: state key_bits are composites packed with the
 * COMMON ranges (must equal the ones passed to every rank's partial). */
int yt_gpu_merge_states_mk(
    const YtPlan* plan, const void* states_device, int64_t state_row_count,
    const uint8_t* col_types,
    const uint64_t* key_zzmin, const uint64_t* key_zzmax,
    const YtExecOptions* options, YtRowset* output, YtStatistics* stats,
    char* errbuf, size_t errlen);

/* STRING-keyed coordinated split (the key shuffle of
 * shuffling_reader.cpp:40-42 applied to string group keys). The bottom
 * query partitions its local string group-by results into YtStateRow
 * records — key_bits = (byte offset within the state's own partition pool
 * slice) << 24 | length — plus the per-partition key-byte pool slices.
 * Partition = splitmix64(FNV-1a(key bytes)) % partition_count; the null
 * key partitions as splitmix64(FNV-basis ^ 0xDEADBEEF12345678) with
 * key_bits 0 and meta bit0 set. The caller exchanges BOTH buffers (counts
 * and byte counts returned per partition); the receiver concatenates its
 * received slices in a fixed segment order and hands the extents to the
 * merge. */
int yt_gpu_query_partial_str(
    const YtPlan* plan, const YtChunk* chunk, const YtExecOptions* options,
    int32_t partition_count, void* states_device, int64_t capacity_rows,
    void* pool_device, int64_t pool_capacity,
    int64_t* part_counts, int64_t* part_pool_bytes,
    YtStatistics* stats, char* errbuf, size_t errlen);

/* string-keyed front merge: states/pool are DEVICE buffers holding the
 * received segments concatenated in order; seg_counts / seg_pool_bytes
 * give each segment's extent (a state's slice-local key reference is
 * rebased by its segment's pool offset). col_types = the original chunk's
 * column types (resolves the sum result type). */
int yt_gpu_merge_states_str(
    const YtPlan* plan, const void* states_device, const int64_t* seg_counts,
    int32_t nseg_in, const void* pool_device, const int64_t* seg_pool_bytes,
    const uint8_t* col_types,
    const YtExecOptions* options, YtRowset* output, YtStatistics* stats,
    char* errbuf, size_t errlen);

/* ---- chunk encoder (host-side product component; the synthetic-data
 * generator — reference writer semantics, integer_column_writer.cpp). ---- */

/* Encodes one int64/uint64 column into reference-format segments.
 * nulls: optional bytemask (1 = null), length n. Segments split every
 * max_segment_values rows (DefaultMaxSegmentValueCount = 128*1024,
 * table_chunk_format/public.h:11). The returned blob is a single allocation;
 * call yt_encoded_column_free. On return, *segments points into the blob. */
typedef struct YtEncodedColumn {
    int32_t segment_count;
    YtSegment* segments;          /* data pointers are HOST pointers into blob */
    void* blob;
    int64_t blob_size;
} YtEncodedColumn;

/* cum_rows_base: rows already written to this column before `values` —
 * feeds the reference's cumulative RowCount_ used in the RLE segment-size
 * estimate (integer_column_writer.cpp:365,371), so a column encoded in
 * parallel slices is bit-identical to a serial encode. Pass 0 normally. */
int yt_encode_int64_column(
    const int64_t* values, const uint8_t* nulls, int64_t n,
    int32_t max_segment_values, int32_t is_unsigned, int64_t cum_rows_base,
    YtEncodedColumn* out, char* errbuf, size_t errlen);

int yt_encode_double_column(
    const double* values, const uint8_t* nulls, int64_t n,
    int32_t max_segment_values,
    YtEncodedColumn* out, char* errbuf, size_t errlen);

/* boolean columns — boolean_column_writer.cpp DumpBooleanValues:
 * [ui64 count][value bitmap, 8-aligned][null bitmap, 8-aligned] */
int yt_encode_bool_column(
    const uint8_t* values, const uint8_t* nulls, int64_t n,
    int32_t max_segment_values,
    YtEncodedColumn* out, char* errbuf, size_t errlen);

void yt_encoded_column_free(YtEncodedColumn* col);

/* Bit-pack primitive, exposed for format tests
 * (core/misc/bit_packed_unsigned_vector-inl.h:34-82). dst must be
 * zero-initialized and hold yt_bitpack_size_words(max_value, n) words.
 * Returns words written. */
int64_t yt_bitpack_size_words(uint64_t max_value, int64_t n);
int64_t yt_bitpack(const uint64_t* values, int64_t n, uint64_t max_value, uint64_t* dst);

#ifdef __cplusplus
} /* extern "C" */
#endif
#endif /* YTQL_GPU_H */
