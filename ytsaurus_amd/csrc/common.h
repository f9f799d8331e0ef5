/* Internal structures shared between the host evaluator and the HIP kernels.
 * Product code (libytql_gpu.so). Not part of the public C-ABI
 * (see /root/repo/include/ytql_gpu.h for the boundary).
 */
#pragma once
#include <stdint.h>

namespace ytql {

/* Resolved device-side segment descriptor. The kernel parses the sub-blob
 * offsets itself from the 8-byte bit-pack headers
 * (reference layout: bit_packed_unsigned_vector-inl.h:34-45 header word =
 * count(56b)|width(8b); segment assembly integer_column_writer.cpp:66-116,
 * 394-489). */
struct DevSeg {
    int32_t type;        /* YT_SEG_* */
    int32_t is_signed;   /* 1 = int64 (zigzag), 0 = uint64; 2 = string column */
    int64_t start_row;   /* first chunk row covered */
    int32_t row_count;
    int32_t col;         /* owning column index */
    uint64_t min_value;
    const uint64_t* blob;
    int64_t blob_bytes;
};

/* Parsed per-segment sub-blob map (built on device by k_parse_segments). */
struct SegEx {
    uint32_t w_values;
    uint32_t w_ids;
    uint32_t w_starts;
    uint32_t run_count;       /* RLE run count / dictionary id count */
    uint32_t dict_size;
    uint32_t flags;
    int64_t off_values_words; /* first data word of the values vector (past header) */
    int64_t off_bitmap_bytes; /* null bitmap (direct formats) */
    int64_t off_ids_words;    /* dictionary ids / RLE ids data words */
    int64_t off_starts_words; /* RLE run-start data words */
    int64_t off_doubles_bytes;
};

/* Postfix expression program; mirrors the oracle's eval_expr semantics
 * (cg_fragment_compiler.cpp arithmetic/relational/logical codegen). */
enum POp : int32_t {
    P_COL = 0, P_LIT_I64 = 1, P_LIT_NULL = 2, P_LIT_DOUBLE = 3,
    P_ADD = 10, P_SUB = 11, P_MUL = 12, P_DIV = 13, P_MOD = 14,
    P_EQ = 20, P_NE = 21, P_LT = 22, P_LE = 23, P_GT = 24, P_GE = 25,
    P_AND = 30, P_OR = 31, P_NOT = 32,
};

struct PInst {
    int32_t op;
    int32_t col;
    uint64_t bits;     /* literal payload */
};

constexpr int kMaxProg = 48;
constexpr int kMaxCols = 8;
constexpr int kMaxAggs = 4;

/* One compiled plan for the device: programs are concatenated postfix
 * streams with (offset,len) per role. */
constexpr int kMaxProj = 8;

constexpr int kMaxPackKeys = 4;

struct DevPlan {
    int32_t ncols;
    uint8_t col_types[kMaxCols];       /* YT_VT_* */
    /* composite packed group key (key_count > 1): each component column's
     * zigzag-space value is biased by kp_base and offset by 1 (0 = null),
     * then packed at kp_shift. Total width <= 62 bits so the table's empty
     * (0x8000...) sentinel can never collide; key_bits == 0 (all-null key)
     * rides the existing side-accumulator path. */
    int32_t kp_count;
    int32_t kp_col[kMaxPackKeys];
    int32_t kp_signed[kMaxPackKeys];
    int32_t kp_shift[kMaxPackKeys];
    int32_t kp_bits[kMaxPackKeys];
    uint64_t kp_base[kMaxPackKeys];
    /* log2(rows per segment) when the column's interior segments are all
     * exactly 1<<shift rows (the writer's 128Ki cap makes this the norm):
     * row→segment is then a shift instead of a binary search. 0 = ragged. */
    int32_t col_uniform_shift[kMaxCols];
    int32_t filter_off, filter_len;    /* -1 len 0 = none */
    int32_t key_off, key_len;          /* key_count==1 only; len 0 = global agg */
    int32_t agg_count;
    int32_t agg_func[kMaxAggs];        /* YT_AGG_* */
    int32_t agg_off[kMaxAggs], agg_len[kMaxAggs];
    int32_t proj_count;                /* scan-without-aggregation mode */
    int32_t proj_off[kMaxProj], proj_len[kMaxProj];
    PInst prog[kMaxProg];
    int32_t prog_len;
};

/* Equi-join device context (one join item, UNIQUE foreign keys — see
 * include/ytql_gpu.h YtJoin). The hash table is open-addressing with the
 * row index as the claim word (hrow -1 = empty); duplicate keys are
 * detected by a post-build verify kernel, so inserts never read other
 * slots' keys (no publish race). Probes run in later kernels only. */
struct JoinDev {
    int32_t active;
    int32_t is_left;
    int32_t pkey_col;             /* primary key column */
    int32_t primary_ncols;        /* P: plan columns >= P are foreign */
    int32_t fval_col[kMaxCols];   /* foreign column per appended slot */
    int32_t f_shift[kMaxCols];    /* uniform row->segment shift per slot */
    int32_t fkey_shift;
    int32_t fkey_col;
    const DevSeg* fsegs;
    const SegEx* fsegex;
    const int32_t* f_off;         /* per FOREIGN column: first segment */
    const int32_t* f_cnt;
    const uint64_t* hkey;
    const int64_t* hrow;          /* -1 = empty (slot claim; dup keys may
                                     occupy several slots — probes stop at
                                     the FIRST matching slot, whose chead
                                     holds the match list) */
    uint64_t hmask;
    int64_t null_row;             /* head of the null-key chain, or -1 */
    int64_t frows;
    /* duplicate foreign keys (registry.cpp MultiJoinOpHelper cross-product
     * expansion): per-slot match-list head + per-foreign-row next links */
    const int64_t* chead;         /* [nslots] chain head row, -1 */
    const int64_t* fnext;         /* [frows] next same-key row, -1 */
    int32_t has_dups;             /* any key (incl. null) with >1 row */
    int32_t pad2_;
};

/* versioned scan-format read: per-segment device
 * descriptor; ts_data/val_data point at device copies of the encoded
 * segment blobs (layouts in include/ytql_gpu.h) */
struct VSegDev {
    int64_t row_start;
    int64_t row_count;
    uint64_t base_timestamp;
    uint32_t exp_w, exp_d, exp_v;
    uint32_t vtype;      /* YT_VSEG_* (include/ytql_gpu.h) */
    uint32_t vflags;     /* YT_VSEG_F_* */
    uint32_t pad_;
    uint64_t base_value;
    const char* ts_data;
    const char* val_data;
};

/* device output value for the scan+project path (16 B) */
struct DevOutVal {
    uint64_t bits;
    uint32_t type;     /* YT_VT_*; YT_VT_NULL for null */
    uint32_t pad_;
};

/* Group hash-table slot layout (generic path):
 *   u64 key_bits | u64 cnt | per-agg { u64 bits, u64 nonnull }
 * stride_u64 = 2 + 2*agg_count. key_bits == 0 means EMPTY; real key 0 and
 * null key live in the side accumulators of TableHdr. */
struct TableHdr {
    uint64_t nslots;          /* power of two */
    uint64_t mask;
    /* side groups: [0] = the in-table empty-sentinel key (bits in
     * side_key_bits[0]), [1] = null key */
    uint64_t side_used[2];    /* 0/1 */
    uint64_t side_key_bits[2];
    uint64_t side_cnt[2];
    uint64_t side_agg[2][2 * kMaxAggs];  /* bits, nonnull per agg */
    unsigned long long ngroups;          /* inserted (excl. side) */
    uint64_t overflow;        /* set when probe failed / group limit hit */
    int64_t group_limit;      /* 0 = unlimited */
};

/* Compacted output record (device→host), generic path */
struct OutGroup {
    uint64_t key_bits;
    uint64_t key_meta;        /* bit0 null-key */
    uint64_t cnt;
    uint64_t agg_bits[kMaxAggs];
    uint64_t agg_nonnull[kMaxAggs];
};

/* Fast-shape descriptor (analysed by the host):
 * filter: optional int64 range on one column; key: direct column or none;
 * aggs: sum(direct col) and/or sum(1). */
struct FastShape {
    int32_t valid;
    int32_t filter_col;       /* -1 = none */
    int64_t filter_lo, filter_hi;   /* inclusive */
    int32_t key_col;          /* -1 = global */
    int32_t nsum;             /* number of sum(col) aggs */
    int32_t sum_col[kMaxAggs];
    int32_t sum_slot[kMaxAggs];     /* agg index of each sum */
    int32_t have_sum1;
};

/* fast-kernel launch descriptors */
struct FastCol {
    int32_t seg_off;      /* into segs/segex: first segment of this column */
    int32_t seg_cnt;
};

struct FastParams {
    int32_t nused;               /* staged columns */
    int32_t tile_rows;
    int32_t tiles_per_seg;       /* uniform by the 128Ki interior-segment cap */
    int32_t filter_idx;          /* index into used cols; -1 none */
    int32_t key_idx;             /* -1 = global aggregate */
    int32_t nsum;
    int32_t sum_idx[kMaxAggs];   /* used-col index of each sum arg */
    int32_t sum_slot[kMaxAggs];  /* agg slot in table layout */
    int32_t agg_count;           /* table stride basis */
    int64_t filter_lo, filter_hi;
    int64_t row_count;
    int32_t nsegs_per_col;
    int32_t ntiles;
    int32_t stage_bm_mask;       /* bit u: used col u has nulls — stage+check */
    int32_t pad2_;
};

/* two-phase partitioned group-by (BASELINE configs 3/4 family):
 * Phase A (k_scan_partition): fused decode+filter, hash-partition rows into
 * kNB buckets as 16-byte {key,val} records (mirrors the reference's
 * in-process shuffle partition, shuffling_reader.cpp:40-42, pushed down to
 * the row level so aggregation can run in LDS).
 * Phase B (k_bucket_agg): one workgroup per bucket aggregates its records
 * in an LDS open-addressing table and emits compacted OutGroups. */
constexpr int kNB = 1024;          /* partition buckets */
constexpr int kHSlots = 2048;      /* LDS table slots per bucket (48 KB -> 3 WGs/CU) */
constexpr uint64_t kEmptyKey = 0x8000000000000000ULL;  /* INT64_MIN bits */

struct PartParams {
    int32_t tile_rows;
    int32_t tiles_per_seg;
    int32_t ntiles;
    int32_t filter_idx;           /* used-col index; -1 none */
    int32_t key_idx;              /* used-col index (required) */
    int32_t val_idx;              /* used-col index of the sum arg; -1 none */
    int32_t nused;
    int32_t has_val_nulls;
    int32_t has_key_nulls;
    int32_t sum_slot;             /* agg slot of the sum; -1 none */
    int32_t agg_count;
    int32_t pad_;
    int64_t filter_lo, filter_hi;
    int64_t bucket_stride;        /* record capacity per bucket */
    int64_t nbucket_stride;       /* null-stream capacity per bucket */
    /* 8-byte packed records: when the key and value zigzag spans fit 64
     * bits together, a record is (kzz - gmin_k) | (vzz - gmin_v) << bits_k
     * — half the partition traffic of {key,val} pairs. */
    int32_t packed_mode;
    int32_t bits_k;
    uint64_t gmin_k, gmin_v;
    int32_t stage_bm_mask;        /* bit u: stage used-col u's null bitmap */
    int32_t has_filter_nulls;
    int32_t stage_val;            /* stage the value column's packed words too */
    /* direct-span mode (small key zigzag span): buckets are key-RANGE
     * slices (bucket = krel >> dshift) and phase B indexes an LDS array
     * directly — no hash probe, no CAS, no stored keys. */
    int32_t direct_mode;
    int32_t dshift;               /* per-bucket key-slot range = 1 << dshift */
    /* LDS bucket-major reorder: pass 2 writes records into an LDS scratch
     * (dense bucket-major within the tile), then a copy phase streams each
     * bucket run to its aligned global claim — every HBM line of the record
     * stream is written whole, by coalesced adjacent-lane stores (the 8B
     * scatter wrote 45 GB for 8 GB of records: partial lines evicted from
     * L2 between stores). Packed records, no val-null stream only. */
    int32_t reorder;
    int32_t store_mode;           /* 0 plain, 1 sc1 write-through, 2 no store (timing floor probe) */
    /* per-WORKGROUP record regions: region = bucket * gridDim + blockIdx,
     * appended locally (LDS running base, init once per WG) — the per-tile
     * global cursor reserve was ~1 atomic per 4 rows (250M global atomics
     * per 1B-row query vs the ~30 G/s chip ceiling = ~8 ms of the pass).
     * cursors[] becomes the per-(bucket,wg) final counts, published at WG
     * end. Requires !has_val_nulls (the null stream keeps 8 XCD subs). */
    int32_t wg_streams;
    /* 64B-aligned record claims: each (tile,bucket) reserves a multiple of 8
     * records and fills the tail with pad records, so every HBM line of the
     * partition stream is written whole by one workgroup within one tile
     * pass (round-1 PMC: unaligned 8B scatter wrote 42 GB for 8 GB of
     * records). Pads: packed records use bit 63; 16B/null-stream records
     * use key == kEmptyKey (real INT64_MIN keys never enter the streams —
     * they are side-slotted). */
    int32_t aligned;
};

/* string-keyed GROUP BY (BASELINE config 5 family): dictionary-encoded
 * string key segments aggregate by per-segment dictionary id into dense
 * per-segment accumulators, then dictionary entries merge across segments
 * in a global table keyed by (hash, byte-exact string compare). */
struct StrGroupParams {
    int32_t key_seg_off;          /* key column segments */
    int32_t key_seg_cnt;
    int32_t val_seg_off;          /* sum-arg column segments; -1 = none */
    int32_t val_seg_cnt;
    int32_t val_is_double;
    int32_t sum_slot;             /* agg slot of the sum; -1 = none */
    int32_t agg_count;
    int32_t tile_rows;
    int32_t tiles_per_seg;
    int32_t ntiles;
    int32_t has_val_nulls;
    int32_t xcd_affine;           /* keep each segment's accumulators in one
                                     XCD's L2 (blockIdx%8 = XCD round-robin) */
    int64_t row_count;
};

/* merge-table slot: rep = (seg+1)<<32 | id (1-based), 0 = empty.
 * The representative's hash is NOT stored: readers recompute it from the
 * precomputed per-entry hash array via (seg, id), which avoids any
 * publication ordering between the claim and a hash field. */
/* 64-byte slot, one cache line: rep claims the slot (CAS); the claimer
 * then publishes the key identity in-slot with RELAXED agent stores —
 * ident = hash48|len16, pfx = the key's first 16 bytes. Readers verify
 * with relaxed loads; a stale (zero / partially visible) identity only
 * demotes the probe to the exact owner-array compare, never a wrong
 * accept: ident mismatch is exact (FNV48+len of equal strings always
 * match), ident match with len <= 16 is decided by pfx (the FULL bytes),
 * len > 16 falls back to the byte compare against the owner's dictionary
 * entry. */
struct StrSlot {
    unsigned long long rep;       /* (seg+1)<<32 | id; 0 = empty */
    uint64_t ident;               /* hash48<<16 | len; 0 = unpublished */
    uint64_t pfx[2];              /* first 16 key bytes, zero-padded */
    uint64_t sum_bits;
    uint64_t cnt;                 /* low 32 = row count, high 32 = nonnull */
    uint64_t pad0_;
    uint64_t pad_;
};
static_assert(sizeof(StrSlot) == 64, "one cache line per slot");

/* ORDER BY ... LIMIT k-selection (see kernels.hip k_topk_*).
 * Range-adaptive digits: candidates are mapped keys in [lo, hi]; digit =
 * (m - lo) >> shift (< 2048 by host choice of shift). Pass 0 also reduces
 * the data min/max so the next pass bins the ACTUAL key range instead of
 * stepping fixed 11-bit prefixes through constant high bits. */
struct TopkPass {
    int32_t order_proj;       /* projection index whose value is the order key */
    int32_t desc;             /* invert mapped key */
    int32_t level0;           /* count nulls + reduce min/max on this pass */
    int32_t shift;
    uint64_t lo, hi;          /* inclusive candidate interval */
};
struct TopkGather {
    int32_t order_proj;
    int32_t desc;
    int32_t all_nonnull;      /* 1 = every non-null row is a strict hit */
    int32_t pad_;
    uint64_t lo, hi;          /* strict: m < lo; tie: lo <= m <= hi */
    int64_t cap_tie;
    int64_t cap_null;
};

/* merged string-state group (device→host) for the string-keyed two-phase
 * front query: counts are full u64 (multi-rank accumulation can exceed the
 * 32-bit halves the single-pass path packs). */
struct OutStrState {
    uint64_t off_len;             /* out_pool offset<<24 | len */
    uint64_t sum_bits;
    uint64_t cnt;
    uint64_t nonnull;
};

/* compacted string group (device→host). 24 B: this array is the bulk of
 * the query's D2H at 100 M groups, so the key ref packs off|len (pool is
 * bounded by the dict blob bytes << 2^48) and counts stay in the slot's
 * cnt|nonnull 32-bit-halves encoding. */
struct OutStrGroup {
    uint64_t off_len;             /* pool_off<<24 | len (len cap 16 MB =
                                     the reference MaxStringValueLength) */
    uint64_t sum_bits;
    uint64_t cnt_nonnull;         /* low 32 = row count, high 32 = nonnull */
};

struct KernelTimes {
    float scan_ms;
    int64_t scan_launches;
    float other_ms;
};

} /* namespace ytql */
