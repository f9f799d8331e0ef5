/* ytql_oracle.c — CPU restatement of the YTsaurus dynamic-table query hot
 * path (scan → filter → group-by → merge), used to pin parity of the GPU
 * executor.
 *
 * TEST INFRASTRUCTURE ONLY. This library is the parity CHECKER and the
 * reported CPU baseline; it is imported only by tests/, __graft_entry__.smoke()
 * and bench.py's cpu_baseline leg. The product path (ytsaurus_amd/libytql_gpu.so)
 * never calls it and has no CPU fallback.
 *
 * Every function cites the reference code (/root/reference, ytsaurus/ytsaurus
 * @2026-08-21) it restates:
 *   - bit-unpack:      yt/yt/core/misc/bit_packed_unsigned_vector-inl.h:108-117,157-186
 *   - int64 decode:    yt/yt/ytlib/table_chunk_format/integer_column_reader.cpp:19-127,391-449
 *   - double decode:   yt/yt/ytlib/table_chunk_format/floating_point_column_reader.cpp:36-60
 *   - null bitmap:     yt/yt/core/misc/bitmap.h (ui8, LSB-first, 1 = null;
 *                      serialization aligned to 8 bytes)
 *   - expressions:     yt/yt/library/query/engine/cg_fragment_compiler.cpp
 *                      (arithmetic :1440-1540 null-propagating, div-by-zero at :2019;
 *                       relational :1601-1720: non-canonical null relations compare
 *                       (rhsIsNull, lhsIsNull) i.e. null < any, result Boolean;
 *                       logical :1547-1599 Kleene)
 *   - group-by:        yt/yt/library/query/engine/cg_routines/registry.cpp:1231-1916
 *                      (InsertGroupRow :1783, InsertIntermediate :1517; flush in
 *                       insert order :1571-1650)
 *   - aggregates:      yt/yt/library/query/engine/udf/sum.c (null-propagating add,
 *                      int64 wraps mod 2^64), min.c, max.c
 *   - merge mode:      cg_fragment_compiler.cpp:4116-4134 (front query re-groups by
 *                      key and Merge(state,state))
 *
 * Parity pinning: the reference binary cannot be built in this container
 * (conanfile.py:20-23 needs bison/m4/ragel/yasm + network) and this path has
 * no Python implementation, so the oracle is pinned against golden vectors
 * transcribed from the reference's own unit tests
 * (yt/yt/library/query/unittests/ql_query_ut.cpp GroupBy family) in
 * tests/golden/ — see tests/test_oracle_golden.py.
 */

#include <stdint.h>
#include <stddef.h>
#include <stdlib.h>
#include <string.h>
#include <stdio.h>
#include <pthread.h>

#include "../include/ytql_gpu.h"   /* shared struct/type declarations only */

#define ORACLE_EXPORT __attribute__((visibility("default")))

/* ------------------------------------------------------------------ */
/* small helpers                                                       */

/* oracle-internal aggregate: count of non-null args (avg's count leg —
 * used by yto_partial to materialize the exact {count,sum} avg state;
 * never part of the public plan surface) */
#define YTO_AGG_NNCNT 100

static void set_err(char* errbuf, size_t errlen, const char* msg)
{
    if (errbuf && errlen) {
        snprintf(errbuf, errlen, "%s", msg);
    }
}

/* ZigZag codec — library/cpp/yt/coding/zig_zag-inl.h:24-35 */
static inline uint64_t zigzag_encode64(int64_t n)
{
    return ((uint64_t)n << 1) ^ (uint64_t)(n >> 63);
}
static inline int64_t zigzag_decode64(uint64_t n)
{
    return (int64_t)((n >> 1) ^ (~(n & 1) + 1));
}

/* ------------------------------------------------------------------ */
/* bit-packed unsigned vector reader                                   */
/* header word = count(low 56b) | width(top 8b)                        */
/* bit_packed_unsigned_vector-inl.h:108-133,157-186                    */

typedef struct {
    const uint64_t* data;   /* first data word (header + 1) */
    uint64_t size;          /* element count */
    unsigned width;
} BitReader;

static BitReader bitreader_init(const void* ptr)
{
    BitReader r;
    uint64_t header = *(const uint64_t*)ptr;
    r.data = (const uint64_t*)ptr + 1;
    r.size = header & ((1ULL << 56) - 1);
    r.width = (unsigned)(header >> 56);
    return r;
}

/* bytes including header — TBitPackedUnsignedVectorReader::GetByteSize */
static int64_t bitreader_byte_size(const BitReader* r)
{
    return (int64_t)(1 + (((uint64_t)r->width * r->size + 63ULL) >> 6)) * 8;
}

/* GetValue — -inl.h:157-178 */
static inline uint64_t bitreader_get(const BitReader* r, uint64_t index)
{
    if (r->width == 0) {
        return 0;
    }
    if (r->width == 64) {
        return r->data[index];
    }
    uint64_t bit_index = index * r->width;
    const uint64_t* word = r->data + (bit_index >> 6);
    unsigned offset = bit_index & 63;
    uint64_t w1 = *word >> offset;
    if (offset + r->width > 64) {
        uint64_t w2 = (word[1] & ((1ULL << ((offset + r->width) & 63)) - 1)) << (64 - offset);
        return w1 | w2;
    }
    return w1 & ((r->width < 64) ? ((1ULL << r->width) - 1) : ~0ULL);
}

/* null bitmap: bit set = null; ui8 LSB-first — core/misc/bitmap.h */
static inline int bitmap_get(const uint8_t* bm, uint64_t index)
{
    return (bm[index >> 3] >> (index & 7)) & 1;
}

static inline int64_t align_up8(int64_t x) { return (x + 7) & ~(int64_t)7; }

int yto_decode_string_column(const YtColumn* col, int64_t row_count,
                             char* out_blob, int64_t blob_cap,
                             int64_t* out_end, uint8_t* nulls);

/* ------------------------------------------------------------------ */
/* segment decode → (int64 bits, null bytemask)                        */

/* Decodes one integer segment into vals (i64 for Int64, raw u64 for Uint64)
 * and nulls bytemask. Returns 0 on success.
 * integer_column_reader.cpp: direct :54-88, dictionary :92-130, RLE via
 * TRleValueExtractorBase (column_reader_detail.h:246-262) + writer run
 * construction integer_column_writer.cpp:394-489. */
static int decode_int_segment(const YtSegment* seg, int is_signed,
                              int64_t* vals, uint8_t* nulls)
{
    const char* ptr = (const char*)seg->data;
    const char* end = ptr + seg->data_size;
    int64_t n = seg->row_count;
    uint64_t minv = seg->min_value;

    switch (seg->type) {
    case YT_SEG_DIRECT_DENSE: {
        BitReader values = bitreader_init(ptr);
        ptr += bitreader_byte_size(&values);
        const uint8_t* nb = (const uint8_t*)ptr;
        ptr += align_up8((n + 7) / 8);
        if ((int64_t)values.size != n || ptr != end) return -1;
        for (int64_t i = 0; i < n; i++) {
            int isnull = bitmap_get(nb, i);
            nulls[i] = (uint8_t)isnull;
            uint64_t data = minv + bitreader_get(&values, i);
            vals[i] = is_signed ? zigzag_decode64(data) : (int64_t)data;
        }
        return 0;
    }
    case YT_SEG_DICTIONARY_DENSE: {
        BitReader dict = bitreader_init(ptr);
        ptr += bitreader_byte_size(&dict);
        BitReader ids = bitreader_init(ptr);
        ptr += bitreader_byte_size(&ids);
        if ((int64_t)ids.size != n || ptr != end) return -1;
        for (int64_t i = 0; i < n; i++) {
            uint64_t id = bitreader_get(&ids, i);
            if (id == 0) {
                nulls[i] = 1;
                vals[i] = 0;
            } else {
                nulls[i] = 0;
                uint64_t data = minv + bitreader_get(&dict, id - 1);
                vals[i] = is_signed ? zigzag_decode64(data) : (int64_t)data;
            }
        }
        return 0;
    }
    case YT_SEG_DIRECT_RLE: {
        BitReader values = bitreader_init(ptr);
        ptr += bitreader_byte_size(&values);
        int64_t run_count = (int64_t)values.size;
        const uint8_t* nb = (const uint8_t*)ptr;
        ptr += align_up8((run_count + 7) / 8);
        BitReader starts = bitreader_init(ptr);
        ptr += bitreader_byte_size(&starts);
        if ((int64_t)starts.size != run_count || ptr != end) return -1;
        int64_t run = 0;
        for (int64_t i = 0; i < n; i++) {
            while (run + 1 < run_count && (int64_t)bitreader_get(&starts, run + 1) <= i) {
                run++;
            }
            int isnull = bitmap_get(nb, run);
            nulls[i] = (uint8_t)isnull;
            uint64_t data = minv + bitreader_get(&values, run);
            vals[i] = is_signed ? zigzag_decode64(data) : (int64_t)data;
        }
        return 0;
    }
    case YT_SEG_DICTIONARY_RLE: {
        BitReader dict = bitreader_init(ptr);
        ptr += bitreader_byte_size(&dict);
        BitReader ids = bitreader_init(ptr);
        ptr += bitreader_byte_size(&ids);
        int64_t run_count = (int64_t)ids.size;
        BitReader starts = bitreader_init(ptr);
        ptr += bitreader_byte_size(&starts);
        if ((int64_t)starts.size != run_count || ptr != end) return -1;
        int64_t run = 0;
        for (int64_t i = 0; i < n; i++) {
            while (run + 1 < run_count && (int64_t)bitreader_get(&starts, run + 1) <= i) {
                run++;
            }
            uint64_t id = bitreader_get(&ids, run);
            if (id == 0) {
                nulls[i] = 1;
                vals[i] = 0;
            } else {
                nulls[i] = 0;
                uint64_t data = minv + bitreader_get(&dict, id - 1);
                vals[i] = is_signed ? zigzag_decode64(data) : (int64_t)data;
            }
        }
        return 0;
    }
    default:
        return -1;
    }
}

/* floating_point_column_reader.cpp:36-60: [u64 count][doubles][null bitmap] */
static int decode_double_segment(const YtSegment* seg, int64_t* vals, uint8_t* nulls)
{
    const char* ptr = (const char*)seg->data;
    const char* end = ptr + seg->data_size;
    uint64_t count = *(const uint64_t*)ptr;
    ptr += 8;
    const double* d = (const double*)ptr;
    ptr += 8 * count;
    const uint8_t* nb = (const uint8_t*)ptr;
    ptr += align_up8(((int64_t)count + 7) / 8);
    if ((int64_t)count != seg->row_count || ptr != end) return -1;
    for (uint64_t i = 0; i < count; i++) {
        nulls[i] = (uint8_t)bitmap_get(nb, i);
        memcpy(&vals[i], &d[i], 8);
    }
    return 0;
}

/* boolean_column_reader.cpp:35-50: [u64 count][value bitmap][null bitmap] */
static int decode_bool_segment(const YtSegment* seg, int64_t* vals, uint8_t* nulls)
{
    const char* ptr = (const char*)seg->data;
    const char* end = ptr + seg->data_size;
    uint64_t count = *(const uint64_t*)ptr;
    ptr += 8;
    const uint8_t* vb = (const uint8_t*)ptr;
    ptr += align_up8(((int64_t)count + 7) / 8);
    const uint8_t* nb = (const uint8_t*)ptr;
    ptr += align_up8(((int64_t)count + 7) / 8);
    if ((int64_t)count != seg->row_count || ptr != end) return -1;
    for (uint64_t i = 0; i < count; i++) {
        nulls[i] = (uint8_t)bitmap_get(nb, i);
        vals[i] = bitmap_get(vb, i);
    }
    return 0;
}

/* Decode a whole column into arrays. vals carries i64 / u64 / double bits. */
ORACLE_EXPORT
int yto_decode_column(const YtColumn* col, int64_t row_count,
                      int64_t* vals, uint8_t* nulls)
{
    int64_t row = 0;
    for (int s = 0; s < col->segment_count; s++) {
        const YtSegment* seg = &col->segments[s];
        int rc;
        if (col->value_type == YT_VT_DOUBLE) {
            rc = decode_double_segment(seg, vals + row, nulls + row);
        } else if (col->value_type == YT_VT_BOOLEAN) {
            rc = decode_bool_segment(seg, vals + row, nulls + row);
        } else {
            rc = decode_int_segment(seg, col->value_type == YT_VT_INT64,
                                    vals + row, nulls + row);
        }
        if (rc != 0) return YT_ERR_INVALID_CHUNK;
        row += seg->row_count;
    }
    return (row == row_count) ? YT_OK : YT_ERR_INVALID_CHUNK;
}

/* ------------------------------------------------------------------ */
/* expression evaluation                                               */

typedef struct {
    uint8_t type;     /* YT_VT_* */
    uint64_t bits;    /* i64/u64/double bits/bool(0/1) */
    const char* str;  /* YT_VT_STRING payload */
    uint32_t len;
} Val;

typedef struct {
    const int64_t* const* col_vals;
    const uint8_t* const* col_nulls;
    const uint8_t* col_types;
    /* string columns: decoded blob + cumulative end offsets (len n+1) */
    const char* const* col_str;
    const int64_t* const* col_str_end;
    int ncols;
    int64_t row;
    const int64_t* rowmap;   /* join expansion: expanded idx -> primary row */
    int jf_base;             /* first joined foreign column (= ncols_primary) */
    int error;        /* YT_ERR_DIV_ZERO etc. */
} EvalCtx;

static Val VNULL(void) { Val v; v.type = YT_VT_NULL; v.bits = 0; v.str = 0; v.len = 0; return v; }

static Val eval_expr(const YtExpr* e, EvalCtx* ctx)
{
    Val v = VNULL();
    switch (e->op) {
    case YT_EX_COLUMN: {
        /* duplicate-key joins expand rows: primary columns read through
         * rowmap (expanded index -> primary row), joined foreign columns
         * (col >= jf_base) are materialized per EXPANDED row */
        int64_t r = (ctx->rowmap && e->col < ctx->jf_base)
            ? ctx->rowmap[ctx->row] : ctx->row;
        if (ctx->col_nulls[e->col][r]) return VNULL();
        v.type = ctx->col_types[e->col];
        if (v.type == YT_VT_STRING) {
            const int64_t* ends = ctx->col_str_end[e->col];
            v.str = ctx->col_str[e->col] + ends[r];
            v.len = (uint32_t)(ends[r + 1] - ends[r]);
            v.bits = 0;
            return v;
        }
        v.bits = (uint64_t)ctx->col_vals[e->col][r];
        return v;
    }
    case YT_EX_LIT_I64:
        v.type = YT_VT_INT64;
        v.bits = (uint64_t)e->lit_i64;
        return v;
    case YT_EX_LIT_DOUBLE:
        v.type = YT_VT_DOUBLE;
        memcpy(&v.bits, &e->lit_dbl, 8);
        return v;
    case YT_EX_LIT_NULL:
        return VNULL();
    case YT_EX_NOT: {
        Val a = eval_expr(e->a, ctx);
        if (a.type == YT_VT_NULL) return VNULL();
        v.type = YT_VT_BOOLEAN;
        v.bits = !a.bits;
        return v;
    }
    default:
        break;
    }

    Val a = eval_expr(e->a, ctx);
    Val b = eval_expr(e->b, ctx);

    if (e->op >= YT_EX_ADD && e->op <= YT_EX_MOD) {
        /* arithmetic: null-propagating (cg_fragment_compiler.cpp arithmetic op) */
        if (a.type == YT_VT_NULL || b.type == YT_VT_NULL) return VNULL();
        if (a.type == YT_VT_DOUBLE) {
            double x, y, r = 0;
            memcpy(&x, &a.bits, 8);
            memcpy(&y, &b.bits, 8);
            switch (e->op) {
            case YT_EX_ADD: r = x + y; break;
            case YT_EX_SUB: r = x - y; break;
            case YT_EX_MUL: r = x * y; break;
            case YT_EX_DIV: r = x / y; break;
            default: ctx->error = YT_ERR_UNSUPPORTED; return VNULL();
            }
            v.type = YT_VT_DOUBLE;
            memcpy(&v.bits, &r, 8);
            return v;
        }
        /* int64/uint64: wrapping two's-complement arithmetic */
        uint64_t x = a.bits, y = b.bits, r = 0;
        int sgn = (a.type == YT_VT_INT64);
        switch (e->op) {
        case YT_EX_ADD: r = x + y; break;
        case YT_EX_SUB: r = x - y; break;
        case YT_EX_MUL: r = x * y; break;
        case YT_EX_DIV:
        case YT_EX_MOD:
            if (y == 0) { ctx->error = YT_ERR_DIV_ZERO; return VNULL(); }
            if (sgn) {
                int64_t sx = (int64_t)x, sy = (int64_t)y;
                if (sx == INT64_MIN && sy == -1) {
                    r = (e->op == YT_EX_DIV) ? (uint64_t)INT64_MIN : 0;
                } else {
                    r = (uint64_t)((e->op == YT_EX_DIV) ? sx / sy : sx % sy);
                }
            } else {
                r = (e->op == YT_EX_DIV) ? x / y : x % y;
            }
            break;
        default: break;
        }
        v.type = a.type;
        v.bits = r;
        return v;
    }

    if (e->op >= YT_EX_EQ && e->op <= YT_EX_GE) {
        /* relational: non-canonical null relations — when either side is
         * null, compare (rhsIsNull, lhsIsNull) unsigned: null < any value,
         * null == null; result is a non-null Boolean
         * (cg_fragment_compiler.cpp:1621-1649). */
        int lt, eq;
        if (a.type == YT_VT_NULL || b.type == YT_VT_NULL) {
            unsigned ln = (a.type == YT_VT_NULL), rn = (b.type == YT_VT_NULL);
            lt = rn < ln;     /* lhs < rhs  <=>  rhsIsNull < lhsIsNull */
            eq = ln == rn;
        } else if (a.type == YT_VT_STRING) {
            uint32_t minlen = a.len < b.len ? a.len : b.len;
            int c = memcmp(a.str, b.str, minlen);
            lt = (c < 0) || (c == 0 && a.len < b.len);
            eq = (c == 0) && (a.len == b.len);
        } else if (a.type == YT_VT_DOUBLE) {
            double x, y;
            memcpy(&x, &a.bits, 8);
            memcpy(&y, &b.bits, 8);
            /* FCmpU*: unordered-or — NaN makes every comparison true in the
             * reference codegen; restate exactly */
            int unordered = (x != x) || (y != y);
            lt = unordered || (x < y);
            eq = unordered || (x == y);
            if (unordered) {
                v.type = YT_VT_BOOLEAN;
                switch (e->op) { /* all unordered comparisons are true */
                case YT_EX_EQ: case YT_EX_NE: case YT_EX_LT:
                case YT_EX_LE: case YT_EX_GT: case YT_EX_GE:
                    v.bits = 1; return v;
                }
            }
        } else if (a.type == YT_VT_INT64) {
            int64_t x = (int64_t)a.bits, y = (int64_t)b.bits;
            lt = x < y;
            eq = x == y;
        } else { /* uint64 / boolean */
            lt = a.bits < b.bits;
            eq = a.bits == b.bits;
        }
        int r = 0;
        switch (e->op) {
        case YT_EX_EQ: r = eq; break;
        case YT_EX_NE: r = !eq; break;
        case YT_EX_LT: r = lt; break;
        case YT_EX_LE: r = lt || eq; break;
        case YT_EX_GT: r = !(lt || eq); break;
        case YT_EX_GE: r = !lt; break;
        }
        v.type = YT_VT_BOOLEAN;
        v.bits = (uint64_t)r;
        return v;
    }

    if (e->op == YT_EX_AND || e->op == YT_EX_OR) {
        /* Kleene logic — cg_fragment_compiler.cpp:1547-1599 */
        int an = (a.type == YT_VT_NULL), bn = (b.type == YT_VT_NULL);
        int av = an ? 0 : (int)(a.bits != 0), bv = bn ? 0 : (int)(b.bits != 0);
        if (e->op == YT_EX_AND) {
            if ((!an && !av) || (!bn && !bv)) { v.type = YT_VT_BOOLEAN; v.bits = 0; return v; }
            if (an || bn) return VNULL();
            v.type = YT_VT_BOOLEAN; v.bits = 1; return v;
        } else {
            if ((!an && av) || (!bn && bv)) { v.type = YT_VT_BOOLEAN; v.bits = 1; return v; }
            if (an || bn) return VNULL();
            v.type = YT_VT_BOOLEAN; v.bits = 0; return v;
        }
    }

    ctx->error = YT_ERR_UNSUPPORTED;
    return VNULL();
}

/* ------------------------------------------------------------------ */
/* group-by hash table (insert-order preserving — flush order mirrors
 * registry.cpp:1571-1650: rows flush in hash-insert order)             */

typedef struct {
    Val* keys;            /* ngroups * key_count */
    Val* states;          /* ngroups * agg_count (sum state: typed or null) */
    uint64_t* acounts;    /* ngroups * agg_count non-null arg counts (avg) */
    uint64_t* rowcounts;  /* rows per group (sum(1) state) */
    int64_t ngroups;
    int64_t cap;
    int64_t* slots;       /* open addressing: index into keys/-1 */
    int64_t nslots;       /* power of 2 */
    int key_count;
    int agg_count;
} GroupTable;

static uint64_t splitmix64(uint64_t x)
{
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

static uint64_t hash_keys(const Val* keys, int key_count)
{
    uint64_t h = 0x12345678ULL;
    for (int i = 0; i < key_count; i++) {
        if (keys[i].type == YT_VT_STRING) {
            uint64_t sh = 0xCBF29CE484222325ULL;
            for (uint32_t k = 0; k < keys[i].len; k++) {
                sh = (sh ^ (uint8_t)keys[i].str[k]) * 0x100000001B3ULL;
            }
            h = splitmix64(h ^ splitmix64(((uint64_t)keys[i].type << 56) ^ sh));
        } else {
            h = splitmix64(h ^ splitmix64(((uint64_t)keys[i].type << 56) ^ keys[i].bits));
        }
    }
    return h;
}

static int keys_eq(const Val* a, const Val* b, int key_count)
{
    for (int i = 0; i < key_count; i++) {
        if (a[i].type != b[i].type) return 0;
        if (a[i].type == YT_VT_STRING) {
            if (a[i].len != b[i].len) return 0;
            if (memcmp(a[i].str, b[i].str, a[i].len) != 0) return 0;
        } else if (a[i].type != YT_VT_NULL && a[i].bits != b[i].bits) {
            return 0;
        }
    }
    return 1;
}

static int gt_init(GroupTable* t, int key_count, int agg_count, int64_t cap_hint)
{
    memset(t, 0, sizeof(*t));
    t->key_count = key_count;
    t->agg_count = agg_count;
    t->cap = cap_hint > 16 ? cap_hint : 16;
    t->nslots = 32;
    while (t->nslots < t->cap * 2) t->nslots <<= 1;
    t->keys = malloc(sizeof(Val) * t->cap * (key_count ? key_count : 1));
    t->states = malloc(sizeof(Val) * t->cap * (agg_count ? agg_count : 1));
    t->acounts = malloc(sizeof(uint64_t) * t->cap * (agg_count ? agg_count : 1));
    t->rowcounts = malloc(sizeof(uint64_t) * t->cap);
    t->slots = malloc(sizeof(int64_t) * t->nslots);
    if (!t->keys || !t->states || !t->acounts || !t->rowcounts || !t->slots) return -1;
    for (int64_t i = 0; i < t->nslots; i++) t->slots[i] = -1;
    return 0;
}

static void gt_free(GroupTable* t)
{
    free(t->keys); free(t->states); free(t->acounts); free(t->rowcounts); free(t->slots);
}

static int gt_grow(GroupTable* t)
{
    int64_t newcap = t->cap * 2;
    int kc = t->key_count ? t->key_count : 1;
    int ac = t->agg_count ? t->agg_count : 1;
    t->keys = realloc(t->keys, sizeof(Val) * newcap * kc);
    t->states = realloc(t->states, sizeof(Val) * newcap * ac);
    t->acounts = realloc(t->acounts, sizeof(uint64_t) * newcap * ac);
    t->rowcounts = realloc(t->rowcounts, sizeof(uint64_t) * newcap);
    if (!t->keys || !t->states || !t->acounts || !t->rowcounts) return -1;
    t->cap = newcap;
    if (t->nslots < newcap * 2) {
        int64_t newslots = t->nslots;
        while (newslots < newcap * 2) newslots <<= 1;
        t->slots = realloc(t->slots, sizeof(int64_t) * newslots);
        if (!t->slots) return -1;
        t->nslots = newslots;
        for (int64_t i = 0; i < t->nslots; i++) t->slots[i] = -1;
        for (int64_t g = 0; g < t->ngroups; g++) {
            uint64_t h = hash_keys(&t->keys[g * t->key_count], t->key_count);
            int64_t s = (int64_t)(h & (uint64_t)(t->nslots - 1));
            while (t->slots[s] != -1) s = (s + 1) & (t->nslots - 1);
            t->slots[s] = g;
        }
    }
    return 0;
}

/* find-or-insert; returns group index, or -1 on OOM */
static int64_t gt_upsert(GroupTable* t, const Val* keys)
{
    uint64_t h = hash_keys(keys, t->key_count);
    int64_t s = (int64_t)(h & (uint64_t)(t->nslots - 1));
    for (;;) {
        int64_t g = t->slots[s];
        if (g == -1) break;
        if (keys_eq(&t->keys[g * t->key_count], keys, t->key_count)) return g;
        s = (s + 1) & (t->nslots - 1);
    }
    if (t->ngroups == t->cap) {
        if (gt_grow(t) != 0) return -1;
        /* slots were rebuilt; re-probe */
        s = (int64_t)(h & (uint64_t)(t->nslots - 1));
        while (t->slots[s] != -1) s = (s + 1) & (t->nslots - 1);
    }
    int64_t g = t->ngroups++;
    t->slots[s] = g;
    memcpy(&t->keys[g * t->key_count], keys, sizeof(Val) * t->key_count);
    for (int a = 0; a < t->agg_count; a++) {
        t->states[g * t->agg_count + a] = VNULL();   /* sum_init — udf/sum.c:3-10 */
        t->acounts[g * t->agg_count + a] = 0;
    }
    t->rowcounts[g] = 0;
    return g;
}

/* sum_update / sum_merge — udf/sum.c:12-65 (null-propagating; int64 wraps) */
static void sum_update_val(Val* state, Val nv)
{
    if (nv.type == YT_VT_NULL) return;
    if (state->type == YT_VT_NULL) { *state = nv; return; }
    if (nv.type == YT_VT_INT64 || nv.type == YT_VT_UINT64) {
        state->bits = state->bits + nv.bits;
    } else if (nv.type == YT_VT_DOUBLE) {
        double x, y;
        memcpy(&x, &state->bits, 8);
        memcpy(&y, &nv.bits, 8);
        x += y;
        memcpy(&state->bits, &x, 8);
    }
}

static void minmax_update_val(Val* state, Val nv, int is_max)
{
    if (nv.type == YT_VT_NULL) return;
    int take = 0;
    if (state->type == YT_VT_NULL) {
        take = 1;
    } else if (nv.type == YT_VT_INT64) {
        int64_t s = (int64_t)state->bits, x = (int64_t)nv.bits;
        take = is_max ? (s < x) : (s >= x);
    } else if (nv.type == YT_VT_UINT64 || nv.type == YT_VT_BOOLEAN) {
        take = is_max ? (state->bits < nv.bits) : (state->bits >= nv.bits);
    } else if (nv.type == YT_VT_DOUBLE) {
        double s, x;
        memcpy(&s, &state->bits, 8);
        memcpy(&x, &nv.bits, 8);
        take = is_max ? (s < x) : (s >= x);
    }
    if (take) *state = nv;
}

/* ------------------------------------------------------------------ */
/* execution                                                           */

typedef struct {
    const YtPlan* plan;
    const YtChunk* chunk;
    int64_t row_begin, row_end;
    int64_t** vals;        /* decoded columns (shared) */
    uint8_t** nulls;
    uint8_t* types;
    char** strblob;
    int64_t** strend;
    int ncols_eff;         /* primary + joined foreign columns */
    const uint8_t* jdrop;  /* INNER-join misses (row excluded pre-filter) */
    const int64_t* rowmap; /* duplicate-key join expansion (or NULL) */
    int jf_base;
    GroupTable table;
    int error;
    int64_t rows_read;
} ScanTask;

static void* scan_worker(void* arg)
{
    ScanTask* t = (ScanTask*)arg;
    const YtPlan* plan = t->plan;
    EvalCtx ctx;
    ctx.col_vals = (const int64_t* const*)t->vals;
    ctx.col_nulls = (const uint8_t* const*)t->nulls;
    ctx.col_types = t->types;
    ctx.col_str = (const char* const*)t->strblob;
    ctx.col_str_end = (const int64_t* const*)t->strend;
    ctx.ncols = t->ncols_eff;
    ctx.rowmap = t->rowmap;
    ctx.jf_base = t->jf_base;
    ctx.error = 0;

    Val keybuf[16];
    int kc = plan->key_count;

    for (int64_t r = t->row_begin; r < t->row_end; r++) {
        ctx.row = r;
        t->rows_read++;
        if (t->jdrop && t->jdrop[r]) continue;
        if (plan->filter) {
            Val f = eval_expr(plan->filter, &ctx);
            if (ctx.error) { t->error = ctx.error; return NULL; }
            if (f.type == YT_VT_NULL || f.bits == 0) continue;
        }
        for (int k = 0; k < kc; k++) {
            keybuf[k] = eval_expr(plan->keys[k], &ctx);
            if (ctx.error) { t->error = ctx.error; return NULL; }
        }
        int64_t g = gt_upsert(&t->table, keybuf);
        if (g < 0) { t->error = YT_ERR_CAPACITY; return NULL; }
        t->table.rowcounts[g]++;
        for (int a = 0; a < plan->agg_count; a++) {
            const YtAgg* agg = plan->aggs[a];
            if (agg->func == YT_AGG_SUM1) continue;  /* rowcounts covers it */
            Val nv = eval_expr(agg->arg, &ctx);
            if (ctx.error) { t->error = ctx.error; return NULL; }
            Val* st = &t->table.states[g * plan->agg_count + a];
            if (agg->func == YT_AGG_SUM) sum_update_val(st, nv);
            else if (agg->func == YT_AGG_AVG) {
                /* avg state {count, arg-typed sum} — profiler avg codegen */
                if (nv.type != YT_VT_NULL) {
                    sum_update_val(st, nv);
                    t->table.acounts[g * plan->agg_count + a]++;
                }
            } else if (agg->func == YT_AGG_FIRST) {
                /* FirstIteration: keep the first non-null in scan order */
                if (st->type == YT_VT_NULL) *st = nv;
            } else if (agg->func == YTO_AGG_NNCNT) {
                if (nv.type != YT_VT_NULL)
                    t->table.acounts[g * plan->agg_count + a]++;
            } else minmax_update_val(st, nv, agg->func == YT_AGG_MAX);
        }
    }
    return NULL;
}

/* write one output row from group g of table */
static int emit_group_row(const YtPlan* plan, GroupTable* t, int64_t g,
                          YtRowset* out)
{
    int kc = plan->key_count, ac = plan->agg_count;
    int ncols = plan->project_count ? plan->project_count : kc + ac;
    if (out->row_count >= out->capacity_rows) return YT_ERR_CAPACITY;

    /* materialize [keys..., finalized aggs...] as a row */
    Val rowvals[32];
    for (int k = 0; k < kc; k++) rowvals[k] = t->keys[g * kc + k];
    for (int a = 0; a < ac; a++) {
        if (plan->aggs[a]->func == YT_AGG_SUM1) {
            Val v; v.type = YT_VT_INT64; v.bits = t->rowcounts[g];
            rowvals[kc + a] = v;
        } else if (plan->aggs[a]->func == YTO_AGG_NNCNT) {
            Val v; v.type = YT_VT_INT64; v.bits = t->acounts[g * ac + a];
            v.str = 0; v.len = 0;
            rowvals[kc + a] = v;
        } else if (plan->aggs[a]->func == YT_AGG_AVG) {
            uint64_t c = t->acounts[g * ac + a];
            Val st = t->states[g * ac + a];
            Val v;
            if (c == 0 || st.type == YT_VT_NULL) {
                v = VNULL();
            } else {
                double sum;
                if (st.type == YT_VT_DOUBLE) memcpy(&sum, &st.bits, 8);
                else if (st.type == YT_VT_UINT64) sum = (double)st.bits;
                else sum = (double)(int64_t)st.bits;
                double r = sum / (double)c;
                v.type = YT_VT_DOUBLE; v.str = 0; v.len = 0;
                memcpy(&v.bits, &r, 8);
            }
            rowvals[kc + a] = v;
        } else {
            rowvals[kc + a] = t->states[g * ac + a];  /* sum/first finalize = copy */
        }
    }

    YtValue* dst = out->values + out->row_count * ncols;
    if (!plan->project_count) {
        /* copy string payloads into the caller's pool */
        for (int i = 0; i < kc + ac; i++) {
            if (rowvals[i].type == YT_VT_STRING) {
                if (out->string_pool_used + rowvals[i].len > out->string_pool_capacity) {
                    return YT_ERR_CAPACITY;
                }
                memcpy(out->string_pool + out->string_pool_used,
                       rowvals[i].str, rowvals[i].len);
                rowvals[i].str = out->string_pool + out->string_pool_used;
                out->string_pool_used += rowvals[i].len;
            }
        }
    }
    if (plan->project_count) {
        /* projection over the group row: column i refers to rowvals[i] */
        int64_t pv[32];
        uint8_t pn[32];
        uint8_t pt[32];
        const int64_t* pvp[32];
        const uint8_t* pnp[32];
        for (int i = 0; i < kc + ac; i++) {
            pv[i] = (int64_t)rowvals[i].bits;
            pn[i] = (rowvals[i].type == YT_VT_NULL);
            pt[i] = pn[i] ? YT_VT_INT64 : rowvals[i].type;
            pvp[i] = &pv[i];
            pnp[i] = &pn[i];
        }
        EvalCtx ctx;
        ctx.col_vals = pvp;
        ctx.col_nulls = pnp;
        ctx.col_types = pt;
        ctx.ncols = kc + ac;
        ctx.row = 0;
        ctx.rowmap = NULL;
        ctx.jf_base = 0;
        ctx.error = 0;
        for (int p = 0; p < plan->project_count; p++) {
            Val v = eval_expr(plan->projects[p], &ctx);
            if (ctx.error) return ctx.error;
            dst[p].id = (uint16_t)p;
            dst[p].type = v.type;
            dst[p].flags = 0;
            dst[p].length = 0;
            dst[p].data.bits = v.bits;
        }
    } else {
        for (int i = 0; i < ncols; i++) {
            dst[i].id = (uint16_t)i;
            dst[i].type = rowvals[i].type;
            dst[i].flags = 0;
            if (rowvals[i].type == YT_VT_STRING) {
                dst[i].length = rowvals[i].len;
                dst[i].data.str = rowvals[i].str;
            } else {
                dst[i].length = 0;
                dst[i].data.bits = rowvals[i].bits;
            }
        }
    }
    out->row_count++;
    out->column_count = ncols;
    return YT_OK;
}

/* ---- ORDER BY ... LIMIT (TTopCollector restatement) ----
 * Comparer mirrors the codegen universal comparer
 * (cg_fragment_compiler.cpp:400-530): null < any, int64 signed,
 * uint64/boolean unsigned, double by value with NaN comparison an error,
 * string memcmp + length tiebreak; descending inverts the per-key outcome.
 * Collector mirrors top_collector-inl.h AddRow: keep the (offset+limit)
 * least rows (strictly-less eviction of the current max), emit sorted
 * ascending from index offset (registry.cpp OrderOpHelper:1948-1997). */
static const YtPlan* g_ord_plan;
static int g_ord_ncols;
static int g_ord_err;

static int cmp_order_vals(const YtValue* a, const YtValue* b)
{
    const YtPlan* p = g_ord_plan;
    for (int i = 0; i < p->order_count; i++) {
        int c = p->order_cols[i];
        const YtValue* x = &a[c];
        const YtValue* y = &b[c];
        int xn = x->type == YT_VT_NULL, yn = y->type == YT_VT_NULL;
        int r = 0;
        if (xn || yn) {
            r = (xn == yn) ? 0 : (xn ? -1 : 1);
        } else if (x->type == YT_VT_DOUBLE) {
            double xv = x->data.dbl, yv = y->data.dbl;
            if (xv != xv || yv != yv) { g_ord_err = 1; return 0; }
            r = xv < yv ? -1 : (xv > yv ? 1 : 0);
        } else if (x->type == YT_VT_STRING) {
            uint32_t lx = x->length, ly = y->length, m = lx < ly ? lx : ly;
            int mc = m ? memcmp(x->data.str, y->data.str, m) : 0;
            r = mc ? (mc < 0 ? -1 : 1) : (lx < ly ? -1 : (lx > ly ? 1 : 0));
        } else if (x->type == YT_VT_INT64) {
            r = x->data.i64 < y->data.i64 ? -1 : (x->data.i64 > y->data.i64 ? 1 : 0);
        } else {  /* UINT64 / BOOLEAN */
            r = x->data.u64 < y->data.u64 ? -1 : (x->data.u64 > y->data.u64 ? 1 : 0);
        }
        if (p->order_desc && p->order_desc[i]) r = -r;
        if (r) return r;
    }
    return 0;
}

static int cmp_order_rowptr(const void* pa, const void* pb)
{
    return cmp_order_vals(*(const YtValue* const*)pa, *(const YtValue* const*)pb);
}

static int ord_validate(const YtPlan* plan, int ncols, char* errbuf, size_t errlen)
{
    if (plan->order_limit <= 0) {
        set_err(errbuf, errlen, "ORDER BY requires LIMIT");
        return YT_ERR_INVALID_PLAN;
    }
    for (int i = 0; i < plan->order_count; i++) {
        if (!plan->order_cols || plan->order_cols[i] < 0 ||
            plan->order_cols[i] >= ncols) {
            set_err(errbuf, errlen, "ORDER BY column out of range");
            return YT_ERR_INVALID_PLAN;
        }
    }
    return YT_OK;
}

/* sort the materialized rowset rows and keep [offset, offset+limit) */
static int apply_order_rowset(const YtPlan* plan, YtRowset* out,
                              char* errbuf, size_t errlen)
{
    int ncols = out->column_count;
    int rc = ord_validate(plan, ncols, errbuf, errlen);
    if (rc != YT_OK) return rc;
    int64_t n = out->row_count;
    const YtValue** idx = malloc(sizeof(YtValue*) * (n ? n : 1));
    YtValue* tmp = malloc(sizeof(YtValue) * (n ? n : 1) * ncols);
    if (!idx || !tmp) { free(idx); free(tmp); return YT_ERR_CAPACITY; }
    for (int64_t i = 0; i < n; i++) idx[i] = out->values + i * ncols;
    g_ord_plan = plan;
    g_ord_ncols = ncols;
    g_ord_err = 0;
    qsort(idx, (size_t)n, sizeof(YtValue*), cmp_order_rowptr);
    if (g_ord_err) {
        free(idx); free(tmp);
        set_err(errbuf, errlen, "NaN in ORDER BY comparison");
        return YT_ERR_LIMIT;
    }
    int64_t b = plan->order_offset < n ? plan->order_offset : n;
    int64_t e = b + plan->order_limit;
    if (e > n) e = n;
    for (int64_t i = b; i < e; i++)
        memcpy(tmp + (i - b) * ncols, idx[i], sizeof(YtValue) * ncols);
    memcpy(out->values, tmp, sizeof(YtValue) * (e - b) * ncols);
    out->row_count = e - b;
    free(idx); free(tmp);
    return YT_OK;
}

/* fold the totals row from materialized [keys..., aggs...] rows */
static void ord_fold_totals(const YtPlan* plan, const YtRowset* out,
                            YtValue* tot)
{
    int kc = plan->key_count, ac = plan->agg_count;
    int ncols = out->column_count;
    for (int a = 0; a < ac; a++) {
        tot[a].id = (uint16_t)(kc + a);
        tot[a].type = YT_VT_NULL;
        tot[a].flags = 0;
        tot[a].length = 0;
        tot[a].data.bits = 0;
    }
    for (int64_t r = 0; r < out->row_count; r++) {
        const YtValue* row = out->values + r * ncols;
        for (int a = 0; a < ac; a++) {
            const YtValue* v = &row[kc + a];
            if (v->type == YT_VT_NULL) continue;
            YtValue* t = &tot[a];
            int fagg = plan->aggs[a]->func;
            if (t->type == YT_VT_NULL) { t->type = v->type; t->data.bits = v->data.bits; continue; }
            if (fagg == YT_AGG_FIRST) {
                /* FirstMerge: keep the first non-null (already in t) */
            } else if (fagg == YT_AGG_SUM || fagg == YT_AGG_SUM1) {
                if (v->type == YT_VT_DOUBLE) t->data.dbl += v->data.dbl;
                else t->data.bits += v->data.bits;   /* mod 2^64, udf/sum.c */
            } else {
                int take;
                if (v->type == YT_VT_DOUBLE)
                    take = fagg == YT_AGG_MAX ? v->data.dbl > t->data.dbl
                                              : v->data.dbl < t->data.dbl;
                else if (v->type == YT_VT_INT64)
                    take = fagg == YT_AGG_MAX ? v->data.i64 > t->data.i64
                                              : v->data.i64 < t->data.i64;
                else
                    take = fagg == YT_AGG_MAX ? v->data.u64 > t->data.u64
                                              : v->data.u64 < t->data.u64;
                if (take) t->data.bits = v->data.bits;
            }
        }
    }
}

static void ord_expr_cols(const YtExpr* e, uint64_t* mask)
{
    if (!e) return;
    if (e->op == YT_EX_COLUMN && e->col >= 0 && e->col < 64)
        *mask |= 1ULL << e->col;
    ord_expr_cols(e->a, mask);
    ord_expr_cols(e->b, mask);
}

/* totals(Before) -> HAVING -> totals(After) -> ORDER/limit -> totals row
 * (folding_profiler.cpp:1810-1815 Process(); parser.ypp:469-481 modes) */
static int finish_group_rowset(const YtPlan* plan, YtRowset* out,
                               char* errbuf, size_t errlen)
{
    int kc = plan->key_count, ac = plan->agg_count;
    int ncols = out->column_count;
    YtValue tot[16];
    int after = plan->totals_mode == 2;
    /* all-null group keys are forbidden on the INTERMEDIATE stream (WITH
     * TOTALS re-folds rows) and in the group-combined-with-order op —
     * registry.cpp ValidateGroupKeyIsNotNull:1460-1476, call sites
     * :1795,:1820; pinned by GroupByWithTotalsNulls ql_query_ut.cpp:3989 */
    if (plan->with_totals || plan->order_count > 0) {
        for (int64_t r = 0; r < out->row_count; r++) {
            const YtValue* row = out->values + r * ncols;
            int allnull = 1;
            for (int k = 0; k < kc; k++)
                if (row[k].type != YT_VT_NULL) { allnull = 0; break; }
            if (allnull && kc > 0) {
                set_err(errbuf, errlen, "Null values are forbidden in group key");
                return YT_ERR_INVALID_PLAN;
            }
        }
    }
    const int64_t had_group_rows = out->row_count;
    if (plan->with_totals && !after) ord_fold_totals(plan, out, tot);
    if (plan->having) {
        uint64_t mask = 0;
        ord_expr_cols(plan->having, &mask);
        int64_t w = 0;
        for (int64_t r = 0; r < out->row_count; r++) {
            YtValue* row = out->values + r * ncols;
            int64_t pv[32];
            uint8_t pn[32], pt[32];
            const int64_t* pvp[32];
            const uint8_t* pnp[32];
            for (int i = 0; i < ncols && i < 32; i++) {
                if ((mask >> i) & 1 && row[i].type == YT_VT_STRING) {
                    set_err(errbuf, errlen,
                            "HAVING over string values: not this round");
                    return YT_ERR_UNSUPPORTED;
                }
                pv[i] = (int64_t)row[i].data.bits;
                pn[i] = row[i].type == YT_VT_NULL;
                pt[i] = pn[i] ? YT_VT_INT64 : row[i].type;
                pvp[i] = &pv[i];
                pnp[i] = &pn[i];
            }
            EvalCtx ctx;
            memset(&ctx, 0, sizeof(ctx));
            ctx.col_vals = pvp;
            ctx.col_nulls = pnp;
            ctx.col_types = pt;
            ctx.ncols = ncols;
            ctx.row = 0;
            ctx.error = 0;
            Val h = eval_expr(plan->having, &ctx);
            if (ctx.error) {
                set_err(errbuf, errlen, "HAVING expression error");
                return ctx.error;
            }
            if (h.type == YT_VT_NULL || h.bits == 0) continue;
            if (w != r)
                memmove(out->values + w * ncols, row, sizeof(YtValue) * ncols);
            w++;
        }
        out->row_count = w;
    }
    if (plan->with_totals && after) ord_fold_totals(plan, out, tot);
    if (plan->order_count > 0) {
        int rc = apply_order_rowset(plan, out, errbuf, errlen);
        if (rc != YT_OK) return rc;
    }
    if (plan->with_totals && had_group_rows > 0) {
        if (out->row_count >= out->capacity_rows) return YT_ERR_CAPACITY;
        YtValue* dst = out->values + out->row_count * ncols;
        for (int k = 0; k < kc; k++) {
            dst[k].id = (uint16_t)k;
            dst[k].type = YT_VT_NULL;
            dst[k].flags = 0;
            dst[k].length = 0;
            dst[k].data.bits = 0;
        }
        for (int a = 0; a < ac; a++) {
            dst[kc + a] = tot[a];
            if (plan->aggs[a]->func == YT_AGG_SUM1 && dst[kc + a].type == YT_VT_NULL) {
                dst[kc + a].type = YT_VT_INT64;   /* sum(1) over zero rows */
                dst[kc + a].data.bits = 0;
            }
        }
        out->row_count++;
        out->totals_row = 1;
    }
    return YT_OK;
}

ORACLE_EXPORT
int yto_execute(const YtPlan* plan, const YtChunk* chunk,
                YtRowset* output, YtStatistics* stats,
                int nthreads, char* errbuf, size_t errlen)
{
    if (nthreads < 1) nthreads = 1;
    for (int a = 0; a < plan->agg_count; a++) {
        int f = plan->aggs[a]->func;
        if ((f < YT_AGG_SUM || f > YT_AGG_AVG) && f != YTO_AGG_NNCNT) {
            set_err(errbuf, errlen, "unknown aggregate function");
            return YT_ERR_UNSUPPORTED;
        }
        if (f == YT_AGG_AVG && plan->with_totals) {
            /* avg-of-avgs is wrong; the reference folds states — refuse */
            set_err(errbuf, errlen, "avg: WITH TOTALS not this round");
            return YT_ERR_UNSUPPORTED;
        }
    }
    int ncols = chunk->column_count;
    int jF = plan->join ? plan->join->foreign_value_count : 0;
    const YtJoin* J2chain = plan->join ? plan->join->next : NULL;
    int jF2 = J2chain ? J2chain->foreign_value_count : 0;
    if (J2chain && J2chain->next) {
        set_err(errbuf, errlen, "join: more than two join items not this round");
        return YT_ERR_UNSUPPORTED;
    }
    int ncols_eff = ncols + jF + jF2;
    uint8_t* jdrop = NULL;
    int64_t n = chunk->row_count;
    int64_t n_scan = chunk->row_count;   /* expanded row count under dup joins */
    int64_t* join_rowmap = NULL;
    int rc = YT_OK;

    int64_t** vals = calloc(ncols_eff, sizeof(int64_t*));
    uint8_t** nulls = calloc(ncols_eff, sizeof(uint8_t*));
    uint8_t* types = calloc(ncols_eff, 1);
    char** strblob = calloc(ncols_eff, sizeof(char*));
    int64_t** strend = calloc(ncols_eff, sizeof(int64_t*));
    for (int c = 0; c < ncols; c++) {
        nulls[c] = malloc(n ? n : 1);
        types[c] = (uint8_t)chunk->columns[c].value_type;
        if (types[c] == YT_VT_STRING) {
            /* size: sum of encoded segment bytes is a lower bound only —
             * decode with growth */
            int64_t cap = 1024;
            for (int s2 = 0; s2 < chunk->columns[c].segment_count; s2++) {
                cap += chunk->columns[c].segments[s2].data_size * 4;
            }
            for (;;) {
                strblob[c] = malloc(cap ? cap : 1);
                strend[c] = malloc(sizeof(int64_t) * (n + 1));
                if (!strblob[c] || !strend[c]) { rc = YT_ERR_CAPACITY; goto done; }
                rc = yto_decode_string_column(&chunk->columns[c], n,
                                              strblob[c], cap, strend[c], nulls[c]);
                if (rc != YT_ERR_CAPACITY) break;
                free(strblob[c]); free(strend[c]);
                strblob[c] = NULL; strend[c] = NULL;
                cap *= 4;
            }
            if (rc != YT_OK) { set_err(errbuf, errlen, "oracle: bad string segment"); goto done; }
            vals[c] = calloc(n ? n : 1, sizeof(int64_t));
            continue;
        }
        vals[c] = malloc(sizeof(int64_t) * (n ? n : 1));
        if (!vals[c]) { rc = YT_ERR_CAPACITY; goto done; }
        rc = yto_decode_column(&chunk->columns[c], n, vals[c], nulls[c]);
        if (rc != YT_OK) { set_err(errbuf, errlen, "oracle: bad segment"); goto done; }
    }

    /* equi-join materialization (registry.cpp MultiJoinOpHelper:599-960):
     * unique foreign keys; joined foreign values become columns
     * [ncols, ncols_eff); INNER misses drop the primary row pre-filter */
    if (plan->join) {
        const YtJoin* J = plan->join;
        const YtChunk* fc = J->foreign;
        {
            int fkt = fc->columns[J->foreign_key_col].value_type;
            int pkt = (J->primary_key_col >= 0 && J->primary_key_col < ncols)
                ? types[J->primary_key_col] : YT_VT_INT64;
            if (fkt == YT_VT_STRING || fkt == YT_VT_DOUBLE ||
                pkt == YT_VT_STRING || pkt == YT_VT_DOUBLE) {
                set_err(errbuf, errlen,
                        "join: int64/uint64/boolean key columns this round");
                rc = YT_ERR_UNSUPPORTED;
                goto done;
            }
        }
        int64_t fn = fc->row_count;
        int64_t* fkey = malloc(sizeof(int64_t) * (fn ? fn : 1));
        uint8_t* fknull = malloc(fn ? fn : 1);
        rc = yto_decode_column(&fc->columns[J->foreign_key_col], fn, fkey, fknull);
        if (rc != YT_OK) { free(fkey); free(fknull);
            set_err(errbuf, errlen, "join: bad foreign key segment"); goto done; }
        int64_t** fvals = calloc(jF, sizeof(int64_t*));
        uint8_t** fnulls = calloc(jF, sizeof(uint8_t*));
        for (int j = 0; j < jF && rc == YT_OK; j++) {
            int cjf = J->foreign_value_cols[j];
            if (fc->columns[cjf].value_type == YT_VT_STRING) {
                set_err(errbuf, errlen, "join: string foreign values not this round");
                rc = YT_ERR_UNSUPPORTED;
                break;
            }
            fvals[j] = malloc(sizeof(int64_t) * (fn ? fn : 1));
            fnulls[j] = malloc(fn ? fn : 1);
            rc = yto_decode_column(&fc->columns[cjf], fn, fvals[j], fnulls[j]);
        }
        /* hash map with per-key CHAIN lists (duplicate foreign keys —
         * registry.cpp MultiJoinOpHelper cross-product expansion):
         * hrow[h] = chain head, hnext[r] = next same-key foreign row */
        uint64_t cap = 2048;
        while (cap < (uint64_t)fn * 2) cap <<= 1;
        int64_t* hrow = NULL;
        uint64_t* hkey = NULL;
        uint8_t* hused = NULL;
        int64_t* hnext = NULL;
        int64_t null_row = -1;
        int has_dups = 0;
        if (rc == YT_OK) {
            hrow = malloc(sizeof(int64_t) * cap);
            hkey = malloc(sizeof(uint64_t) * cap);
            hused = calloc(cap, 1);
            hnext = malloc(sizeof(int64_t) * (fn ? fn : 1));
            for (int64_t r2 = 0; r2 < fn; r2++) hnext[r2] = -1;
            for (int64_t r2 = 0; r2 < fn && rc == YT_OK; r2++) {
                if (fknull[r2]) {
                    if (null_row >= 0) has_dups = 1;
                    hnext[r2] = null_row;
                    null_row = r2;
                    continue;
                }
                uint64_t h = splitmix64((uint64_t)fkey[r2]) & (cap - 1);
                for (;;) {
                    if (!hused[h]) { hused[h] = 1; hkey[h] = (uint64_t)fkey[r2]; hrow[h] = r2; break; }
                    if (hkey[h] == (uint64_t)fkey[r2]) {
                        has_dups = 1;
                        hnext[r2] = hrow[h];
                        hrow[h] = r2;
                        break;
                    }
                    h = (h + 1) & (cap - 1);
                }
            }
        }
        if (rc == YT_OK && has_dups &&
            (plan->order_count > 0 || plan->agg_count == 0)) {
            set_err(errbuf, errlen,
                    "join: duplicate foreign keys with ORDER BY / plain scan "
                    "not this round (GROUP BY plans supported)");
            rc = YT_ERR_UNSUPPORTED;
        }
        if (rc == YT_OK && !has_dups) {
            /* unique keys: per-primary-row foreign columns (fast path) */
            for (int j = 0; j < jF; j++) {
                vals[ncols + j] = calloc(n ? n : 1, sizeof(int64_t));
                nulls[ncols + j] = malloc(n ? n : 1);
                memset(nulls[ncols + j], 1, n ? n : 1);
                types[ncols + j] = (uint8_t)fc->columns[J->foreign_value_cols[j]].value_type;
            }
            jdrop = calloc(n ? n : 1, 1);
            int pk = J->primary_key_col;
            for (int64_t r2 = 0; r2 < n; r2++) {
                int64_t frow = -1;
                if (nulls[pk][r2]) {
                    frow = null_row;      /* null joins null (eq-comparer) */
                } else {
                    uint64_t h = splitmix64((uint64_t)vals[pk][r2]) & (cap - 1);
                    while (hused[h]) {
                        if (hkey[h] == (uint64_t)vals[pk][r2]) { frow = hrow[h]; break; }
                        h = (h + 1) & (cap - 1);
                    }
                }
                if (frow < 0) {
                    if (!J->is_left) jdrop[r2] = 1;
                    continue;             /* LEFT: joined cols stay null */
                }
                for (int j = 0; j < jF; j++) {
                    if (!fnulls[j][frow]) {
                        vals[ncols + j][r2] = fvals[j][frow];
                        nulls[ncols + j][r2] = 0;
                    }
                }
            }
        } else if (rc == YT_OK) {
            /* duplicate keys: expand (primary, match) pairs; primary
             * columns read through rowmap, foreign columns materialized
             * per EXPANDED row */
            int64_t exp_cap = n ? n : 1;
            int64_t n_exp = 0;
            int64_t* eprim = malloc(sizeof(int64_t) * exp_cap);
            int64_t* ematch = malloc(sizeof(int64_t) * exp_cap);
            int pk = J->primary_key_col;
            for (int64_t r2 = 0; r2 < n && rc == YT_OK; r2++) {
                int64_t frow = -1;
                if (nulls[pk][r2]) {
                    frow = null_row;
                } else {
                    uint64_t h = splitmix64((uint64_t)vals[pk][r2]) & (cap - 1);
                    while (hused[h]) {
                        if (hkey[h] == (uint64_t)vals[pk][r2]) { frow = hrow[h]; break; }
                        h = (h + 1) & (cap - 1);
                    }
                }
                if (frow < 0 && !J->is_left) continue;
                do {
                    if (n_exp == exp_cap) {
                        exp_cap *= 2;
                        eprim = realloc(eprim, sizeof(int64_t) * exp_cap);
                        ematch = realloc(ematch, sizeof(int64_t) * exp_cap);
                        if (!eprim || !ematch) { rc = YT_ERR_CAPACITY; break; }
                    }
                    eprim[n_exp] = r2;
                    ematch[n_exp] = frow;
                    n_exp++;
                    frow = frow >= 0 ? hnext[frow] : -1;
                } while (frow >= 0);
            }
            if (rc == YT_OK) {
                for (int j = 0; j < jF; j++) {
                    vals[ncols + j] = calloc(n_exp ? n_exp : 1, sizeof(int64_t));
                    nulls[ncols + j] = malloc(n_exp ? n_exp : 1);
                    memset(nulls[ncols + j], 1, n_exp ? n_exp : 1);
                    types[ncols + j] = (uint8_t)fc->columns[J->foreign_value_cols[j]].value_type;
                    for (int64_t e2 = 0; e2 < n_exp; e2++) {
                        int64_t m2 = ematch[e2];
                        if (m2 >= 0 && !fnulls[j][m2]) {
                            vals[ncols + j][e2] = fvals[j][m2];
                            nulls[ncols + j][e2] = 0;
                        }
                    }
                }
                join_rowmap = eprim;
                eprim = NULL;
                n_scan = n_exp;
            }
            free(eprim); free(ematch);
        }
        free(fkey); free(fknull);
        for (int j = 0; j < jF; j++) { if (fvals) free(fvals[j]); if (fnulls) free(fnulls[j]); }
        free(fvals); free(fnulls); free(hrow); free(hkey); free(hused); free(hnext);
        if (rc != YT_OK) goto done;
    }

    /* second join item (snowflake chain): joins the row AS EXTENDED by item
     * 0 — its key may be one of item 0's appended columns. UNIQUE foreign
     * keys only (the cross-product machinery binds item 0). Columns
     * [ncols+jF, ncols_eff) are indexed by SCAN row like item 0's. */
    if (J2chain) {
        const YtJoin* J2 = J2chain;
        const YtChunk* fc = J2->foreign;
        int base2 = ncols + jF;
        int kp = J2->primary_key_col;
        if (kp < 0 || kp >= base2) {
            set_err(errbuf, errlen, "join: key column out of range");
            rc = YT_ERR_INVALID_PLAN;
            goto done;
        }
        if (types[kp] == YT_VT_STRING || types[kp] == YT_VT_DOUBLE) {
            set_err(errbuf, errlen,
                    "join: int64/uint64/boolean key columns this round");
            rc = YT_ERR_UNSUPPORTED;
            goto done;
        }
        int64_t fn = fc->row_count;
        int64_t* fkey = malloc(sizeof(int64_t) * (fn ? fn : 1));
        uint8_t* fknull = malloc(fn ? fn : 1);
        rc = yto_decode_column(&fc->columns[J2->foreign_key_col], fn, fkey, fknull);
        if (rc != YT_OK) { free(fkey); free(fknull);
            set_err(errbuf, errlen, "join: bad foreign key segment"); goto done; }
        int64_t** fvals = calloc(jF2, sizeof(int64_t*));
        uint8_t** fnulls = calloc(jF2, sizeof(uint8_t*));
        for (int j = 0; j < jF2 && rc == YT_OK; j++) {
            int cjf = J2->foreign_value_cols[j];
            if (fc->columns[cjf].value_type == YT_VT_STRING) {
                set_err(errbuf, errlen, "join: string foreign values not this round");
                rc = YT_ERR_UNSUPPORTED;
                break;
            }
            fvals[j] = malloc(sizeof(int64_t) * (fn ? fn : 1));
            fnulls[j] = malloc(fn ? fn : 1);
            rc = yto_decode_column(&fc->columns[cjf], fn, fvals[j], fnulls[j]);
        }
        uint64_t cap = 2048;
        while (cap < (uint64_t)fn * 2) cap <<= 1;
        int64_t* hrow = NULL;
        uint64_t* hkey = NULL;
        uint8_t* hused = NULL;
        int64_t null_row = -1;
        if (rc == YT_OK) {
            hrow = malloc(sizeof(int64_t) * cap);
            hkey = malloc(sizeof(uint64_t) * cap);
            hused = calloc(cap, 1);
            for (int64_t r2 = 0; r2 < fn && rc == YT_OK; r2++) {
                if (fknull[r2]) {
                    if (null_row >= 0) {
                        set_err(errbuf, errlen,
                                "join: duplicate foreign keys on a non-first "
                                "join item not this round");
                        rc = YT_ERR_UNSUPPORTED;
                    }
                    null_row = r2;
                    continue;
                }
                uint64_t h = splitmix64((uint64_t)fkey[r2]) & (cap - 1);
                for (;;) {
                    if (!hused[h]) { hused[h] = 1; hkey[h] = (uint64_t)fkey[r2]; hrow[h] = r2; break; }
                    if (hkey[h] == (uint64_t)fkey[r2]) {
                        set_err(errbuf, errlen,
                                "join: duplicate foreign keys on a non-first "
                                "join item not this round");
                        rc = YT_ERR_UNSUPPORTED;
                        break;
                    }
                    h = (h + 1) & (cap - 1);
                }
            }
        }
        if (rc == YT_OK) {
            for (int j = 0; j < jF2; j++) {
                vals[base2 + j] = calloc(n_scan ? n_scan : 1, sizeof(int64_t));
                nulls[base2 + j] = malloc(n_scan ? n_scan : 1);
                memset(nulls[base2 + j], 1, n_scan ? n_scan : 1);
                types[base2 + j] = (uint8_t)fc->columns[J2->foreign_value_cols[j]].value_type;
            }
            if (!jdrop) jdrop = calloc(n_scan ? n_scan : 1, 1);
            for (int64_t e2 = 0; e2 < n_scan; e2++) {
                if (jdrop[e2]) continue;
                int64_t kr = (join_rowmap && kp < ncols) ? join_rowmap[e2] : e2;
                int64_t frow = -1;
                if (nulls[kp][kr]) {
                    frow = null_row;
                } else {
                    uint64_t h = splitmix64((uint64_t)vals[kp][kr]) & (cap - 1);
                    while (hused[h]) {
                        if (hkey[h] == (uint64_t)vals[kp][kr]) { frow = hrow[h]; break; }
                        h = (h + 1) & (cap - 1);
                    }
                }
                if (frow < 0) {
                    if (!J2->is_left) jdrop[e2] = 1;
                    continue;
                }
                for (int j = 0; j < jF2; j++) {
                    if (!fnulls[j][frow]) {
                        vals[base2 + j][e2] = fvals[j][frow];
                        nulls[base2 + j][e2] = 0;
                    }
                }
            }
        }
        free(fkey); free(fknull);
        for (int j = 0; j < jF2; j++) { if (fvals) free(fvals[j]); if (fnulls) free(fnulls[j]); }
        free(fvals); free(fnulls); free(hrow); free(hkey); free(hused);
        if (rc != YT_OK) goto done;
    }

    output->row_count = 0;
    output->string_pool_used = 0;
    output->totals_row = 0;
    if ((plan->with_totals || plan->having) && plan->key_count == 0) {
        set_err(errbuf, errlen, "WITH TOTALS / HAVING requires GROUP BY");
        rc = YT_ERR_INVALID_PLAN;
        goto done;
    }
    if ((plan->with_totals || plan->having) && plan->project_count) {
        set_err(errbuf, errlen, "WITH TOTALS / HAVING with projections: not this round");
        rc = YT_ERR_UNSUPPORTED;
        goto done;
    }

    if (plan->agg_count == 0 && plan->key_count == 0) {
        /* plain scan+filter+project */
        EvalCtx ctx;
        ctx.col_vals = (const int64_t* const*)vals;
        ctx.col_nulls = (const uint8_t* const*)nulls;
        ctx.col_types = types;
        ctx.col_str = (const char* const*)strblob;
        ctx.col_str_end = (const int64_t* const*)strend;
        ctx.ncols = ncols_eff;
        ctx.rowmap = NULL;
        ctx.jf_base = 0;
        ctx.error = 0;
        int np = plan->project_count;
        for (int64_t r = 0; r < n; r++) {
            ctx.row = r;
            if (jdrop && jdrop[r]) continue;
            if (plan->filter) {
                Val f = eval_expr(plan->filter, &ctx);
                if (ctx.error) { rc = ctx.error; set_err(errbuf, errlen, "expr error"); goto done; }
                if (f.type == YT_VT_NULL || f.bits == 0) continue;
            }
            if (output->row_count >= output->capacity_rows) { rc = YT_ERR_CAPACITY; goto done; }
            YtValue* dst = output->values + output->row_count * np;
            for (int p = 0; p < np; p++) {
                Val v = eval_expr(plan->projects[p], &ctx);
                if (ctx.error) { rc = ctx.error; set_err(errbuf, errlen, "expr error"); goto done; }
                dst[p].id = (uint16_t)p;
                dst[p].type = v.type;
                dst[p].flags = 0;
                dst[p].length = 0;
                dst[p].data.bits = v.bits;
            }
            output->row_count++;
        }
        output->column_count = np;
        if (plan->order_count > 0) {
            rc = apply_order_rowset(plan, output, errbuf, errlen);
            if (rc != YT_OK) goto done;
        }
        if (stats) {
            stats->rows_read = n;
            stats->rows_written = output->row_count;
            stats->grouped_row_count = 0;
        }
        goto done;
    }

    /* group-by: partition rows across threads on segment-ish boundaries,
     * per-thread tables, then merge in thread order (mirrors the reference's
     * per-tablet bottom queries + front merge, executor.cpp:761) */
    {
        ScanTask* tasks = calloc(nthreads, sizeof(ScanTask));
        pthread_t* tids = malloc(sizeof(pthread_t) * nthreads);
        int64_t per = (n_scan + nthreads - 1) / nthreads;
        int actual = 0;
        for (int i = 0; i < nthreads; i++) {
            int64_t b = (int64_t)i * per;
            int64_t e = b + per > n_scan ? n_scan : b + per;
            if (b >= e) break;
            tasks[actual].plan = plan;
            tasks[actual].chunk = chunk;
            tasks[actual].row_begin = b;
            tasks[actual].row_end = e;
            tasks[actual].vals = vals;
            tasks[actual].nulls = nulls;
            tasks[actual].types = types;
            tasks[actual].strblob = strblob;
            tasks[actual].strend = strend;
            tasks[actual].ncols_eff = ncols_eff;
            tasks[actual].jdrop = jdrop;
            tasks[actual].rowmap = join_rowmap;
            tasks[actual].jf_base = ncols;
            gt_init(&tasks[actual].table, plan->key_count, plan->agg_count, 1024);
            actual++;
        }
        if (actual == 1) {
            scan_worker(&tasks[0]);
        } else {
            for (int i = 0; i < actual; i++) pthread_create(&tids[i], NULL, scan_worker, &tasks[i]);
            for (int i = 0; i < actual; i++) pthread_join(tids[i], NULL);
        }
        for (int i = 0; i < actual; i++) {
            if (tasks[i].error) { rc = tasks[i].error; set_err(errbuf, errlen, "scan error"); }
        }
        if (rc == YT_OK) {
            GroupTable* final_t;
            GroupTable merged;
            if (actual == 1) {
                final_t = &tasks[0].table;
            } else {
                gt_init(&merged, plan->key_count, plan->agg_count, 1024);
                for (int i = 0; i < actual; i++) {
                    GroupTable* pt = &tasks[i].table;
                    for (int64_t g = 0; g < pt->ngroups; g++) {
                        int64_t mg = gt_upsert(&merged, &pt->keys[g * pt->key_count]);
                        merged.rowcounts[mg] += pt->rowcounts[g];
                        for (int a = 0; a < plan->agg_count; a++) {
                            Val st = pt->states[g * plan->agg_count + a];
                            Val* dst = &merged.states[mg * plan->agg_count + a];
                            if (plan->aggs[a]->func == YT_AGG_SUM) sum_update_val(dst, st);
                            else if (plan->aggs[a]->func == YT_AGG_AVG) {
                                sum_update_val(dst, st);
                                merged.acounts[mg * plan->agg_count + a] +=
                                    pt->acounts[g * plan->agg_count + a];
                            } else if (plan->aggs[a]->func == YT_AGG_FIRST) {
                                /* FirstMerge: tasks iterate in row order */
                                if (dst->type == YT_VT_NULL) *dst = st;
                            } else if (plan->aggs[a]->func != YT_AGG_SUM1)
                                minmax_update_val(dst, st, plan->aggs[a]->func == YT_AGG_MAX);
                        }
                    }
                }
                final_t = &merged;
            }
            for (int64_t g = 0; g < final_t->ngroups && rc == YT_OK; g++) {
                rc = emit_group_row(plan, final_t, g, output);
            }
            output->column_count = plan->project_count
                ? plan->project_count : plan->key_count + plan->agg_count;
            if (rc == YT_OK && (plan->order_count > 0 || plan->with_totals ||
                                plan->having)) {
                rc = finish_group_rowset(plan, output, errbuf, errlen);
            }
            if (stats) {
                stats->rows_read = n;
                stats->rows_written = output->row_count;
                stats->grouped_row_count = final_t->ngroups;
            }
            if (actual != 1) gt_free(&merged);
        }
        for (int i = 0; i < actual; i++) gt_free(&tasks[i].table);
        free(tasks);
        free(tids);
    }

done:
    for (int c = 0; c < ncols_eff; c++) {
        if (vals) free(vals[c]);
        if (nulls) free(nulls[c]);
        if (strblob) free(strblob[c]);
        if (strend) free(strend[c]);
    }
    free(vals); free(nulls); free(types); free(strblob); free(strend);
    free(jdrop);
    free(join_rowmap);
    return rc;
}

/* ------------------------------------------------------------------ */
/* two-phase path: partial states + partition, and merge               */
/* (restates the bottom/front split + in-process shuffle:               */
/*  engine/coordinator.cpp:420-505, engine_api/shuffling_reader.cpp:21-88) */

/* Partition hash over the group key — internal choice (need not match the
 * reference's farm fingerprint, results are order-free); MUST match the GPU
 * library's yt_partition_hash. */
ORACLE_EXPORT
uint64_t yto_partition_hash(uint64_t key_bits, int key_is_null)
{
    return splitmix64(key_bits ^ (key_is_null ? 0xDEADBEEF12345678ULL : 0));
}

/* Bottom query for the round-1 state family: 1 int64/bool key, aggs =
 * {sum(expr), sum(1)} in any order. Emits YtStateRow records grouped into
 * partition_count buckets (bucket-major). */
ORACLE_EXPORT
int yto_partial(const YtPlan* plan, const YtChunk* chunk,
                int32_t partition_count,
                YtStateRow* out, int64_t capacity_rows,
                int64_t* part_counts,
                int nthreads, char* errbuf, size_t errlen)
{
    /* join at the bottom query: the local group-by below delegates to
     * yto_execute, whose join path (incl. dup-key expansion) applies before
     * grouping — matching coordinator.cpp:130-170 keeping the JoinClause in
     * the bottom query. Multi-key + join stays refused (yto_partial_mk). */
    if (plan->key_count != 1) { set_err(errbuf, errlen, "partial: need 1 key"); return YT_ERR_UNSUPPORTED; }
    int sum_idx = -1;
    int is_avg = 0;
    for (int a = 0; a < plan->agg_count; a++) {
        int f = plan->aggs[a]->func;
        if (f == YT_AGG_SUM || f == YT_AGG_AVG || f == YT_AGG_MIN ||
            f == YT_AGG_MAX || f == YT_AGG_FIRST) {
            if (sum_idx >= 0) { set_err(errbuf, errlen, "partial: one value-carrying agg max this round"); return YT_ERR_UNSUPPORTED; }
            sum_idx = a;
            is_avg = (f == YT_AGG_AVG);
        }
        else if (f != YT_AGG_SUM1) { set_err(errbuf, errlen, "partial: sum/avg/min/max/first/sum1 only"); return YT_ERR_UNSUPPORTED; }
    }

    /* run the local group-by via yto_execute on a plan without projection.
     * avg(x) lowers to the {count,sum} state the reference's coordinated
     * avg carries (GroupByWithAvgCoordinated ql_query_ut.cpp:2760): the
     * local avg slot becomes sum(x), and a second pass with the filter
     * extended by (x == x) — null comparison yields null, which the WHERE
     * clause drops — counts the non-null args per group exactly. */
    YtPlan local = *plan;
    local.project_count = 0;
    local.projects = NULL;
    YtAgg local_aggs[17];
    const YtAgg* local_agg_ptrs[17];
    if (is_avg) {
        if (plan->agg_count > 16) { set_err(errbuf, errlen, "partial: too many aggregates"); return YT_ERR_UNSUPPORTED; }
        for (int a = 0; a < plan->agg_count; a++) {
            local_aggs[a] = *plan->aggs[a];
            if (a == sum_idx) local_aggs[a].func = YT_AGG_SUM;
            local_agg_ptrs[a] = &local_aggs[a];
        }
        /* the avg count leg: exact non-null arg count per group */
        local_aggs[plan->agg_count].func = YTO_AGG_NNCNT;
        local_aggs[plan->agg_count].arg = plan->aggs[sum_idx]->arg;
        local_agg_ptrs[plan->agg_count] = &local_aggs[plan->agg_count];
        local.agg_count = plan->agg_count + 1;
        local.aggs = local_agg_ptrs;
    }

    int64_t cap = capacity_rows;
    YtValue* tmp = malloc(sizeof(YtValue) * cap * (1 + plan->agg_count));
    if (!tmp) return YT_ERR_CAPACITY;
    YtRowset rs;
    memset(&rs, 0, sizeof(rs));
    rs.values = tmp;
    rs.capacity_rows = cap;
    YtStatistics st;
    memset(&st, 0, sizeof(st));
    int rc = yto_execute(&local, chunk, &rs, &st, nthreads, errbuf, errlen);
    if (rc != YT_OK) { free(tmp); return rc; }

    /* To know nonnull counts we recompute: a state with non-null sum means
     * nonnull >= 1; exact nonnull count is not needed for sum-merge parity —
     * only null-ness matters (sum.c:12-22). Encode nonnull = 1 for non-null
     * state, 0 for null state. */
    int ncols = 1 + local.agg_count;   /* incl. the internal count leg */

    int64_t* counts = calloc(partition_count, sizeof(int64_t));
    for (int64_t r = 0; r < rs.row_count; r++) {
        const YtValue* row = rs.values + r * ncols;
        int knull = (row[0].type == YT_VT_NULL);
        counts[yto_partition_hash(row[0].data.bits, knull) % (uint64_t)partition_count]++;
    }
    int64_t total = 0;
    int64_t* offs = calloc(partition_count, sizeof(int64_t));
    for (int p = 0; p < partition_count; p++) { offs[p] = total; total += counts[p]; }
    if (total > capacity_rows) { free(tmp); free(counts); free(offs); return YT_ERR_CAPACITY; }

    for (int64_t r = 0; r < rs.row_count; r++) {
        const YtValue* row = rs.values + r * ncols;
        int knull = (row[0].type == YT_VT_NULL);
        int64_t p = (int64_t)(yto_partition_hash(row[0].data.bits, knull) % (uint64_t)partition_count);
        YtStateRow* sr = &out[offs[p]++];
        sr->key_bits = row[0].data.bits;
        uint64_t rowcount = 0, sum_bits = 0, nonnull = 0;
        for (int a = 0; a < plan->agg_count; a++) {
            if (plan->aggs[a]->func == YT_AGG_SUM1) rowcount = row[1 + a].data.bits;
            else if (a == sum_idx) {
                sum_bits = row[1 + a].data.bits;
                nonnull = (row[1 + a].type != YT_VT_NULL);
            }
        }
        if (is_avg)
            nonnull = row[1 + plan->agg_count].data.bits;   /* count leg */
        uint64_t sum_dbl = 0;
        if (sum_idx >= 0 && row[1 + sum_idx].type == YT_VT_DOUBLE) sum_dbl = 2;
        sr->meta = (uint64_t)knull | sum_dbl | (nonnull << 8);
        sr->sum_bits = sum_bits;
        sr->row_count = rowcount;
    }
    for (int p = 0; p < partition_count; p++) part_counts[p] = counts[p];
    free(tmp); free(counts); free(offs);
    return YT_OK;
}

/* Front-query merge + finalize over YtStateRow records.
 * Output columns follow the plan: [key, aggs...] or projection. */
ORACLE_EXPORT
int yto_merge(const YtPlan* plan, const YtStateRow* states, int64_t nstates,
              YtRowset* output, char* errbuf, size_t errlen)
{
    if (plan->key_count != 1) { set_err(errbuf, errlen, "merge: need 1 key"); return YT_ERR_UNSUPPORTED; }
    GroupTable t;
    gt_init(&t, 1, plan->agg_count, 1024);
    int key_type = YT_VT_INT64;
    /* key static type: a boolean-valued key expr yields boolean outputs */
    for (int64_t i = 0; i < nstates; i++) {
        Val key;
        if (states[i].meta & 1) {
            key = VNULL();
        } else {
            key.type = (uint8_t)key_type;
            key.bits = states[i].key_bits;
        }
        int64_t g = gt_upsert(&t, &key);
        t.rowcounts[g] += states[i].row_count;
        uint64_t nonnull = states[i].meta >> 8;
        for (int a = 0; a < plan->agg_count; a++) {
            int f = plan->aggs[a]->func;
            if (f != YT_AGG_SUM && f != YT_AGG_AVG && f != YT_AGG_MIN &&
                f != YT_AGG_MAX && f != YT_AGG_FIRST) continue;
            if (nonnull) {
                Val nv;
                nv.type = (states[i].meta & 2) ? YT_VT_DOUBLE : YT_VT_INT64;
                nv.bits = states[i].sum_bits;
                nv.str = 0; nv.len = 0;
                Val* st2 = &t.states[g * plan->agg_count + a];
                if (f == YT_AGG_FIRST) {
                    if (st2->type == YT_VT_NULL) *st2 = nv;
                } else if (f == YT_AGG_MIN || f == YT_AGG_MAX)
                    minmax_update_val(st2, nv, f == YT_AGG_MAX);
                else
                    sum_update_val(st2, nv);
                /* avg finalize divides by the accumulated non-null count */
                t.acounts[g * plan->agg_count + a] += nonnull;
            }
        }
    }
    output->row_count = 0;
    output->totals_row = 0;
    int rc = YT_OK;
    for (int64_t g = 0; g < t.ngroups && rc == YT_OK; g++) {
        rc = emit_group_row(plan, &t, g, output);
    }
    /* ORDER BY / WITH TOTALS / HAVING apply at the front (coordinator)
     * query, exactly like the GPU merge path */
    if (rc == YT_OK && (plan->order_count > 0 || plan->with_totals ||
                        plan->having)) {
        output->column_count = 1 + plan->agg_count;
        rc = finish_group_rowset(plan, output, errbuf, errlen);
    }
    gt_free(&t);
    return rc;
}

/* bit-unpack KAT helper: unpack a packed vector (with header) into out */
ORACLE_EXPORT
int64_t yto_bitunpack(const void* packed, uint64_t* out, int64_t max_out)
{
    BitReader r = bitreader_init(packed);
    int64_t n = (int64_t)r.size < max_out ? (int64_t)r.size : max_out;
    for (int64_t i = 0; i < n; i++) out[i] = bitreader_get(&r, i);
    return (int64_t)r.size;
}

/* ------------------------------------------------------------------ */
/* string segment decode — string_column_reader.cpp:
 *   GetOffset :39-42 (offset(i) = expected_length*(i+1) + ZigZagDecode32(packed[i])),
 *   SetStringValue :44-66, dictionary extract :77-124, direct extract :130-168,
 *   RLE via TRleStringValueExtractorBase; blob layouts mirror
 *   string_column_writer.cpp dumps (ids/offsets/data orders per dump fn).
 * YtSegment.min_value carries expected_length.                         */

static inline int32_t zigzag_decode32(uint32_t n)
{
    return (int32_t)((n >> 1) ^ (~(n & 1) + 1));
}

static uint64_t str_offset(const BitReader* offsets, uint64_t expected, int64_t i)
{
    if (i < 0) return 0;
    return expected * (uint64_t)(i + 1)
         + (uint64_t)(int64_t)zigzag_decode32((uint32_t)bitreader_get(offsets, i));
}

/* decode one string segment; appends string bytes to out_blob (cursor
 * *blob_used), writes per-row cumulative ends via out_end[row+1] and nulls. */
static int decode_string_segment(const YtSegment* seg,
                                 char* out_blob, int64_t blob_cap, int64_t* blob_used,
                                 int64_t* out_end, uint8_t* nulls)
{
    const char* ptr = (const char*)seg->data;
    int64_t n = seg->row_count;
    uint64_t expected = seg->min_value;

    BitReader starts = {0}, ids = {0}, offs = {0};
    const uint8_t* nb = NULL;
    const char* data = NULL;
    int64_t run_count = 0;
    int is_dict = 0, is_rle = 0;

    switch (seg->type) {
    case YT_SEG_DIRECT_DENSE: {
        offs = bitreader_init(ptr);
        ptr += bitreader_byte_size(&offs);
        nb = (const uint8_t*)ptr;
        ptr += align_up8((n + 7) / 8);
        data = ptr;
        if ((int64_t)offs.size != n) return -1;
        break;
    }
    case YT_SEG_DICTIONARY_DENSE: {
        ids = bitreader_init(ptr);
        ptr += bitreader_byte_size(&ids);
        offs = bitreader_init(ptr);
        ptr += bitreader_byte_size(&offs);
        data = ptr;
        is_dict = 1;
        if ((int64_t)ids.size != n) return -1;
        break;
    }
    case YT_SEG_DIRECT_RLE: {
        starts = bitreader_init(ptr);
        ptr += bitreader_byte_size(&starts);
        run_count = (int64_t)starts.size;
        offs = bitreader_init(ptr);
        ptr += bitreader_byte_size(&offs);
        nb = (const uint8_t*)ptr;
        ptr += align_up8((run_count + 7) / 8);
        data = ptr;
        is_rle = 1;
        break;
    }
    case YT_SEG_DICTIONARY_RLE: {
        starts = bitreader_init(ptr);
        ptr += bitreader_byte_size(&starts);
        run_count = (int64_t)starts.size;
        ids = bitreader_init(ptr);
        ptr += bitreader_byte_size(&ids);
        offs = bitreader_init(ptr);
        ptr += bitreader_byte_size(&offs);
        data = ptr;
        is_rle = 1;
        is_dict = 1;
        break;
    }
    default:
        return -1;
    }

    int64_t run = 0;
    for (int64_t i = 0; i < n; i++) {
        int64_t idx = i;
        if (is_rle) {
            while (run + 1 < run_count && (int64_t)bitreader_get(&starts, run + 1) <= i) run++;
            idx = run;
        }
        int isnull;
        int64_t sbeg = 0, slen = 0;
        if (is_dict) {
            uint64_t id = bitreader_get(&ids, idx);
            isnull = (id == 0);
            if (!isnull) {
                sbeg = (int64_t)str_offset(&offs, expected, (int64_t)id - 2);
                slen = (int64_t)str_offset(&offs, expected, (int64_t)id - 1) - sbeg;
            }
        } else {
            isnull = bitmap_get(nb, idx);
            if (!isnull) {
                sbeg = (int64_t)str_offset(&offs, expected, idx - 1);
                slen = (int64_t)str_offset(&offs, expected, idx) - sbeg;
            }
        }
        nulls[i] = (uint8_t)isnull;
        if (!isnull) {
            if (*blob_used + slen > blob_cap) return -2;
            memcpy(out_blob + *blob_used, data + sbeg, slen);
            *blob_used += slen;
        }
        out_end[i + 1] = *blob_used;
    }
    return 0;
}

/* ---- multi-key two-phase (common-range composite packing) ----
 * Mirrors the GPU path exactly (evaluator pack_group_key + DevPlan kp_*):
 * component i (plain int64/uint64/boolean column) is coded as
 *   0                      when null
 *   1 + (zzspace(v) - lo)  otherwise   (zigzag for int64, raw bits else)
 * packed at a running shift; total width <= 62 bits. The caller passes the
 * CROSS-RANK-reduced [lo, hi] per component so every rank and the merge
 * pack identically. */
typedef struct {
    int count;
    int col[4];
    int is_signed[4];
    int shift[4];
    int bits[4];
    uint64_t base[4];
} MkPack;

static int mk_resolve(const YtPlan* plan, const uint8_t* col_types,
                      const uint64_t* kzmin, const uint64_t* kzmax,
                      MkPack* mk, char* errbuf, size_t errlen)
{
    mk->count = plan->key_count;
    int shift = 0;
    for (int i = 0; i < plan->key_count; i++) {
        const YtExpr* e = plan->keys[i];
        if (!e || e->op != YT_EX_COLUMN) {
            set_err(errbuf, errlen, "mk: plain key columns only");
            return YT_ERR_UNSUPPORTED;
        }
        int c = e->col;
        mk->col[i] = c;
        uint8_t vt = col_types ? col_types[c] : YT_VT_INT64;
        mk->is_signed[i] = vt == YT_VT_INT64;
        uint64_t lo = kzmin[i], hi = kzmax[i];
        uint64_t span = hi >= lo ? hi - lo : 0;
        int bits = 1;
        while (bits < 64 && ((span + 1) >> bits) != 0) bits++;
        mk->base[i] = lo;
        mk->bits[i] = bits;
        mk->shift[i] = shift;
        shift += bits;
    }
    if (shift > 62) {
        set_err(errbuf, errlen, "mk: composite key wider than 62 bits");
        return YT_ERR_UNSUPPORTED;
    }
    return YT_OK;
}

static uint64_t mk_pack_row(const MkPack* mk, const YtValue* row)
{
    uint64_t kb = 0;
    for (int i = 0; i < mk->count; i++) {
        uint64_t enc = 0;
        if (row[i].type != YT_VT_NULL) {
            uint64_t z = mk->is_signed[i]
                ? zigzag_encode64((int64_t)row[i].data.bits)
                : row[i].data.bits;
            enc = 1 + (z - mk->base[i]);
        }
        kb |= enc << mk->shift[i];
    }
    return kb;
}

ORACLE_EXPORT
int yto_partial_mk(const YtPlan* plan, const YtChunk* chunk,
                   int32_t partition_count,
                   const uint64_t* key_zzmin, const uint64_t* key_zzmax,
                   YtStateRow* out, int64_t capacity_rows,
                   int64_t* part_counts,
                   int nthreads, char* errbuf, size_t errlen)
{
    if (plan->join) { set_err(errbuf, errlen, "partial: join with multi-key GROUP BY not this round"); return YT_ERR_UNSUPPORTED; }
    if (plan->key_count < 1 || plan->key_count > 4) { set_err(errbuf, errlen, "partial: 1..4 keys"); return YT_ERR_UNSUPPORTED; }
    int sum_idx = -1;
    int is_avg = 0;
    for (int a = 0; a < plan->agg_count; a++) {
        int f = plan->aggs[a]->func;
        if (f == YT_AGG_SUM || f == YT_AGG_AVG || f == YT_AGG_MIN ||
            f == YT_AGG_MAX || f == YT_AGG_FIRST) {
            if (sum_idx >= 0) { set_err(errbuf, errlen, "partial: one value-carrying agg max this round"); return YT_ERR_UNSUPPORTED; }
            sum_idx = a;
            is_avg = (f == YT_AGG_AVG);
        }
        else if (f != YT_AGG_SUM1) { set_err(errbuf, errlen, "partial: sum/avg/min/max/first/sum1 only"); return YT_ERR_UNSUPPORTED; }
    }
    uint8_t ct[64];
    memset(ct, YT_VT_INT64, sizeof(ct));
    for (int c = 0; c < chunk->column_count && c < 64; c++)
        ct[c] = (uint8_t)chunk->columns[c].value_type;
    MkPack mk;
    int rc = mk_resolve(plan, ct, key_zzmin, key_zzmax, &mk, errbuf, errlen);
    if (rc != YT_OK) return rc;

    YtPlan local = *plan;
    local.project_count = 0;
    local.projects = NULL;
    YtAgg local_aggs[17];
    const YtAgg* local_agg_ptrs[17];
    if (is_avg) {
        if (plan->agg_count > 16) { set_err(errbuf, errlen, "partial: too many aggregates"); return YT_ERR_UNSUPPORTED; }
        for (int a = 0; a < plan->agg_count; a++) {
            local_aggs[a] = *plan->aggs[a];
            if (a == sum_idx) local_aggs[a].func = YT_AGG_SUM;
            local_agg_ptrs[a] = &local_aggs[a];
        }
        local_aggs[plan->agg_count].func = YTO_AGG_NNCNT;
        local_aggs[plan->agg_count].arg = plan->aggs[sum_idx]->arg;
        local_agg_ptrs[plan->agg_count] = &local_aggs[plan->agg_count];
        local.agg_count = plan->agg_count + 1;
        local.aggs = local_agg_ptrs;
    }
    int kc = plan->key_count;
    int ncols = kc + local.agg_count;
    int64_t cap = capacity_rows;
    YtValue* tmp = malloc(sizeof(YtValue) * cap * ncols);
    if (!tmp) { set_err(errbuf, errlen, "partial_mk: oom"); return YT_ERR_CAPACITY; }
    YtRowset rs;
    memset(&rs, 0, sizeof(rs));
    rs.values = tmp;
    rs.capacity_rows = cap;
    YtStatistics st;
    memset(&st, 0, sizeof(st));
    rc = yto_execute(&local, chunk, &rs, &st, nthreads, errbuf, errlen);
    if (rc != YT_OK) {
        if (rc == YT_ERR_CAPACITY) set_err(errbuf, errlen, "partial_mk: local group rowset too small");
        free(tmp);
        return rc;
    }

    int64_t* counts = calloc(partition_count, sizeof(int64_t));
    for (int64_t r = 0; r < rs.row_count; r++) {
        uint64_t kb = mk_pack_row(&mk, rs.values + r * ncols);
        counts[yto_partition_hash(kb, 0) % (uint64_t)partition_count]++;
    }
    int64_t total = 0;
    int64_t* offs = calloc(partition_count, sizeof(int64_t));
    for (int p = 0; p < partition_count; p++) { offs[p] = total; total += counts[p]; }
    if (total > capacity_rows) { free(tmp); free(counts); free(offs); set_err(errbuf, errlen, "partial_mk: state buffer too small"); return YT_ERR_CAPACITY; }
    for (int64_t r = 0; r < rs.row_count; r++) {
        const YtValue* row = rs.values + r * ncols;
        uint64_t kb = mk_pack_row(&mk, row);
        int64_t p = (int64_t)(yto_partition_hash(kb, 0) % (uint64_t)partition_count);
        YtStateRow* sr = &out[offs[p]++];
        sr->key_bits = kb;
        uint64_t rowcount = 0, sum_bits = 0, nonnull = 0, sum_dbl = 0;
        for (int a = 0; a < plan->agg_count; a++) {
            if (plan->aggs[a]->func == YT_AGG_SUM1) rowcount = row[kc + a].data.bits;
            else if (a == sum_idx) {
                sum_bits = row[kc + a].data.bits;
                nonnull = (row[kc + a].type != YT_VT_NULL);
                if (row[kc + a].type == YT_VT_DOUBLE) sum_dbl = 2;
            }
        }
        if (is_avg)
            nonnull = row[kc + plan->agg_count].data.bits;   /* count leg */
        sr->meta = sum_dbl | (nonnull << 8);
        sr->sum_bits = sum_bits;
        sr->row_count = rowcount;
    }
    for (int p = 0; p < partition_count; p++) part_counts[p] = counts[p];
    free(tmp); free(counts); free(offs);
    return YT_OK;
}

ORACLE_EXPORT
int yto_merge_mk(const YtPlan* plan, const YtStateRow* states, int64_t nstates,
                 const uint8_t* col_types,
                 const uint64_t* key_zzmin, const uint64_t* key_zzmax,
                 YtRowset* output, char* errbuf, size_t errlen)
{
    if (plan->key_count < 1 || plan->key_count > 4) { set_err(errbuf, errlen, "merge: 1..4 keys"); return YT_ERR_UNSUPPORTED; }
    MkPack mk;
    int rc = mk_resolve(plan, col_types, key_zzmin, key_zzmax, &mk, errbuf, errlen);
    if (rc != YT_OK) return rc;
    GroupTable t;
    gt_init(&t, 1, plan->agg_count, 1024);
    for (int64_t i = 0; i < nstates; i++) {
        Val key;
        key.type = YT_VT_UINT64;
        key.bits = states[i].key_bits;
        key.str = 0; key.len = 0;
        int64_t g = gt_upsert(&t, &key);
        if (g < 0) { gt_free(&t); return YT_ERR_CAPACITY; }
        t.rowcounts[g] += states[i].row_count;
        uint64_t nonnull = states[i].meta >> 8;
        for (int a = 0; a < plan->agg_count; a++) {
            int f = plan->aggs[a]->func;
            if (f != YT_AGG_SUM && f != YT_AGG_AVG && f != YT_AGG_MIN &&
                f != YT_AGG_MAX && f != YT_AGG_FIRST) continue;
            if (nonnull) {
                Val nv;
                nv.type = (states[i].meta & 2) ? YT_VT_DOUBLE : YT_VT_INT64;
                nv.bits = states[i].sum_bits;
                nv.str = 0; nv.len = 0;
                Val* st2 = &t.states[g * plan->agg_count + a];
                if (f == YT_AGG_FIRST) {
                    if (st2->type == YT_VT_NULL) *st2 = nv;
                } else if (f == YT_AGG_MIN || f == YT_AGG_MAX)
                    minmax_update_val(st2, nv, f == YT_AGG_MAX);
                else
                    sum_update_val(st2, nv);
                t.acounts[g * plan->agg_count + a] += nonnull;
            }
        }
    }
    output->row_count = 0;
    output->totals_row = 0;
    int ncols = plan->key_count + plan->agg_count;
    output->column_count = ncols;
    for (int64_t g = 0; g < t.ngroups; g++) {
        if (output->row_count >= output->capacity_rows) { gt_free(&t); return YT_ERR_CAPACITY; }
        YtValue* dst = output->values + output->row_count * ncols;
        uint64_t kb = t.keys[g].bits;
        for (int i = 0; i < mk.count; i++) {
            uint64_t mask = mk.bits[i] >= 64 ? ~0ULL : ((1ULL << mk.bits[i]) - 1);
            uint64_t code = (kb >> mk.shift[i]) & mask;
            dst[i].id = (uint16_t)i;
            dst[i].flags = 0;
            dst[i].length = 0;
            if (code == 0) {
                dst[i].type = YT_VT_NULL;
                dst[i].data.bits = 0;
            } else {
                uint64_t z = code - 1 + mk.base[i];
                dst[i].type = col_types ? col_types[mk.col[i]] : YT_VT_INT64;
                dst[i].data.bits = mk.is_signed[i]
                    ? (uint64_t)zigzag_decode64(z) : z;
            }
        }
        for (int a = 0; a < plan->agg_count; a++) {
            YtValue* v = &dst[mk.count + a];
            v->id = (uint16_t)(mk.count + a);
            v->flags = 0;
            v->length = 0;
            if (plan->aggs[a]->func == YT_AGG_SUM1) {
                v->type = YT_VT_INT64;
                v->data.bits = t.rowcounts[g];
            } else if (plan->aggs[a]->func == YT_AGG_AVG) {
                uint64_t c2 = t.acounts[g * plan->agg_count + a];
                Val st2 = t.states[g * plan->agg_count + a];
                if (c2 == 0 || st2.type == YT_VT_NULL) {
                    v->type = YT_VT_NULL;
                    v->data.bits = 0;
                } else {
                    double sum2;
                    if (st2.type == YT_VT_DOUBLE) memcpy(&sum2, &st2.bits, 8);
                    else sum2 = (double)(int64_t)st2.bits;
                    double rr = sum2 / (double)c2;
                    v->type = YT_VT_DOUBLE;
                    memcpy(&v->data.bits, &rr, 8);
                }
            } else {
                Val st2 = t.states[g * plan->agg_count + a];
                v->type = st2.type;
                v->data.bits = st2.bits;
            }
        }
        output->row_count++;
    }
    gt_free(&t);
    return YT_OK;
}

/* ---- string-keyed two-phase (shuffling_reader.cpp key shuffle applied to
 * string group keys). State rows reuse YtStateRow with key_bits =
 * (offset within the state's own partition pool slice)<<24 | len; the
 * null key has key_bits 0 and meta bit0. Partition =
 * splitmix64(FNV-1a(key bytes)) % world; null key partitions as
 * splitmix64(FNV-basis ^ 0xDEADBEEF12345678). Must match the GPU
 * k_strst_* kernels bit-for-bit. */
static uint64_t fnv1a_c(const char* p, uint32_t len)
{
    uint64_t h = 0xCBF29CE484222325ULL;
    for (uint32_t k = 0; k < len; k++)
        h = (h ^ (uint8_t)p[k]) * 0x100000001B3ULL;
    return h;
}

ORACLE_EXPORT
int yto_partial_str(const YtPlan* plan, const YtChunk* chunk,
                    int32_t partition_count,
                    YtStateRow* out, int64_t capacity_rows,
                    char* pool_out, int64_t pool_capacity,
                    int64_t* part_counts, int64_t* part_pool_bytes,
                    int nthreads, char* errbuf, size_t errlen)
{
    if (plan->key_count != 1) { set_err(errbuf, errlen, "partial_str: need 1 key"); return YT_ERR_UNSUPPORTED; }
    int sum_idx = -1;
    for (int a = 0; a < plan->agg_count; a++) {
        if (plan->aggs[a]->func == YT_AGG_SUM) {
            if (sum_idx >= 0) { set_err(errbuf, errlen, "partial_str: one sum agg max"); return YT_ERR_UNSUPPORTED; }
            sum_idx = a;
        } else if (plan->aggs[a]->func != YT_AGG_SUM1) {
            set_err(errbuf, errlen, "partial_str: sum/sum1 only");
            return YT_ERR_UNSUPPORTED;
        }
    }
    YtPlan local = *plan;
    local.project_count = 0;
    local.projects = NULL;

    int64_t cap = chunk->row_count + 16;
    int ncols = 1 + plan->agg_count;
    YtValue* tmp = malloc(sizeof(YtValue) * cap * ncols);
    char* tpool = malloc((size_t)(pool_capacity > 0 ? pool_capacity : 1));
    if (!tmp || !tpool) { free(tmp); free(tpool); return YT_ERR_CAPACITY; }
    YtRowset rs;
    memset(&rs, 0, sizeof(rs));
    rs.values = tmp;
    rs.capacity_rows = cap;
    rs.string_pool = tpool;
    rs.string_pool_capacity = pool_capacity;
    YtStatistics st;
    memset(&st, 0, sizeof(st));
    int rc = yto_execute(&local, chunk, &rs, &st, nthreads, errbuf, errlen);
    if (rc != YT_OK) { free(tmp); free(tpool); return rc; }

    const int p_null = (int)(splitmix64(
        0xCBF29CE484222325ULL ^ 0xDEADBEEF12345678ULL) % (uint64_t)partition_count);
    int64_t* counts = calloc(partition_count, sizeof(int64_t));
    int64_t* bytes = calloc(partition_count, sizeof(int64_t));
    for (int64_t r = 0; r < rs.row_count; r++) {
        const YtValue* row = rs.values + r * ncols;
        if (row[0].type == YT_VT_NULL) { counts[p_null]++; continue; }
        uint64_t h = splitmix64(fnv1a_c(row[0].data.str, row[0].length));
        int p = (int)(h % (uint64_t)partition_count);
        counts[p]++;
        bytes[p] += row[0].length;
    }
    int64_t rtot = 0, btot = 0;
    int64_t* roff = calloc(partition_count, sizeof(int64_t));
    int64_t* boff = calloc(partition_count, sizeof(int64_t));
    int64_t* bbase = calloc(partition_count, sizeof(int64_t));
    for (int p = 0; p < partition_count; p++) {
        roff[p] = rtot;
        bbase[p] = btot;
        rtot += counts[p];
        btot += bytes[p];
    }
    if (rtot > capacity_rows || btot > pool_capacity) {
        free(tmp); free(tpool); free(counts); free(bytes);
        free(roff); free(boff); free(bbase);
        set_err(errbuf, errlen, "partial_str: state/pool buffer too small");
        return YT_ERR_CAPACITY;
    }
    for (int64_t r = 0; r < rs.row_count; r++) {
        const YtValue* row = rs.values + r * ncols;
        int knull = (row[0].type == YT_VT_NULL);
        int p;
        uint64_t key_bits = 0;
        if (knull) {
            p = p_null;
        } else {
            uint64_t h = splitmix64(fnv1a_c(row[0].data.str, row[0].length));
            p = (int)(h % (uint64_t)partition_count);
            int64_t bo = boff[p];
            memcpy(pool_out + bbase[p] + bo, row[0].data.str, row[0].length);
            key_bits = ((uint64_t)bo << 24) | row[0].length;
            boff[p] += row[0].length;
        }
        YtStateRow* sr = &out[roff[p]++];
        uint64_t rowcount = 0, sum_bits = 0, nonnull = 0, sum_dbl = 0;
        for (int a = 0; a < plan->agg_count; a++) {
            if (plan->aggs[a]->func == YT_AGG_SUM1) rowcount = row[1 + a].data.bits;
            else if (a == sum_idx) {
                sum_bits = row[1 + a].data.bits;
                nonnull = (row[1 + a].type != YT_VT_NULL);
                if (row[1 + a].type == YT_VT_DOUBLE) sum_dbl = 2;
            }
        }
        sr->key_bits = key_bits;
        sr->meta = (uint64_t)knull | sum_dbl | (nonnull << 8);
        sr->sum_bits = sum_bits;
        sr->row_count = rowcount;
    }
    for (int p = 0; p < partition_count; p++) {
        part_counts[p] = counts[p];
        part_pool_bytes[p] = bytes[p];
    }
    free(tmp); free(tpool); free(counts); free(bytes);
    free(roff); free(boff); free(bbase);
    return YT_OK;
}

ORACLE_EXPORT
int yto_merge_str(const YtPlan* plan, const YtStateRow* states,
                  const int64_t* seg_counts, int nseg,
                  const char* pool, const int64_t* seg_pool_bytes,
                  YtRowset* output, char* errbuf, size_t errlen)
{
    if (plan->key_count != 1) { set_err(errbuf, errlen, "merge_str: need 1 key"); return YT_ERR_UNSUPPORTED; }
    int sum_idx = -1;
    for (int a = 0; a < plan->agg_count; a++) {
        if (plan->aggs[a]->func == YT_AGG_SUM) sum_idx = a;
    }
    int64_t n = 0;
    for (int s = 0; s < nseg; s++) n += seg_counts[s];
    /* open-addressing table keyed by string bytes */
    uint64_t nslots = 64;
    while (nslots < (uint64_t)n * 2 + 8) nslots <<= 1;
    typedef struct {
        const char* p;         /* NULL = empty */
        uint32_t len;
        uint64_t cnt, sum_bits, nonnull, sum_dbl;
    } MSlot;
    MSlot* slots = calloc(nslots, sizeof(MSlot));
    MSlot nullg;
    memset(&nullg, 0, sizeof(nullg));
    int has_null = 0;
    int64_t at = 0, ngroups = 0;
    for (int s = 0; s < nseg; s++) {
        int64_t pbase = 0;
        for (int s2 = 0; s2 < s; s2++) pbase += seg_pool_bytes[s2];
        for (int64_t i = 0; i < seg_counts[s]; i++, at++) {
            const YtStateRow* sr = &states[at];
            uint64_t nonnull = sr->meta >> 8;
            if (sr->meta & 1) {
                has_null = 1;
                nullg.cnt += sr->row_count;
                if (nonnull) {
                    if (sr->meta & 2) {
                        double d;
                        memcpy(&d, &nullg.sum_bits, 8);
                        double d2;
                        uint64_t b2 = sr->sum_bits;
                        memcpy(&d2, &b2, 8);
                        d += d2;
                        memcpy(&nullg.sum_bits, &d, 8);
                        nullg.sum_dbl = 1;
                    } else {
                        nullg.sum_bits += sr->sum_bits;
                    }
                    nullg.nonnull += nonnull;
                }
                continue;
            }
            const char* kp = pool + pbase + (int64_t)(sr->key_bits >> 24);
            uint32_t klen = (uint32_t)(sr->key_bits & 0xFFFFFF);
            uint64_t h = splitmix64(fnv1a_c(kp, klen));
            uint64_t idx = h & (nslots - 1);
            for (;;) {
                MSlot* sl = &slots[idx];
                if (!sl->p) {
                    sl->p = kp;
                    sl->len = klen;
                    ngroups++;
                    /* fall through to accumulate */
                }
                if (sl->len == klen && memcmp(sl->p, kp, klen) == 0) {
                    sl->cnt += sr->row_count;
                    if (nonnull) {
                        if (sr->meta & 2) {
                            double d;
                            memcpy(&d, &sl->sum_bits, 8);
                            double d2;
                            uint64_t b2 = sr->sum_bits;
                            memcpy(&d2, &b2, 8);
                            d += d2;
                            memcpy(&sl->sum_bits, &d, 8);
                            sl->sum_dbl = 1;
                        } else {
                            sl->sum_bits += sr->sum_bits;
                        }
                        sl->nonnull += nonnull;
                    }
                    break;
                }
                idx = (idx + 1) & (nslots - 1);
            }
        }
    }
    output->row_count = 0;
    output->string_pool_used = 0;
    output->column_count = 1 + plan->agg_count;
    if (ngroups + has_null > output->capacity_rows) {
        free(slots);
        set_err(errbuf, errlen, "merge_str: output rowset too small");
        return YT_ERR_CAPACITY;
    }
    int ncols = 1 + plan->agg_count;
    for (uint64_t i = 0; i <= nslots; i++) {
        const MSlot* sl;
        int is_null_row = (i == nslots);
        if (is_null_row) {
            if (!has_null) break;
            sl = &nullg;
        } else {
            sl = &slots[i];
            if (!sl->p) continue;
        }
        YtValue* dst = output->values + output->row_count * ncols;
        dst[0].id = 0;
        dst[0].flags = 0;
        if (is_null_row) {
            dst[0].type = YT_VT_NULL;
            dst[0].length = 0;
            dst[0].data.bits = 0;
        } else {
            if (output->string_pool_used + sl->len >
                (uint64_t)output->string_pool_capacity) {
                free(slots);
                set_err(errbuf, errlen, "merge_str: string pool too small");
                return YT_ERR_CAPACITY;
            }
            char* dstp = output->string_pool + output->string_pool_used;
            memcpy(dstp, sl->p, sl->len);
            dst[0].type = YT_VT_STRING;
            dst[0].length = sl->len;
            dst[0].data.str = dstp;
            output->string_pool_used += sl->len;
        }
        for (int a = 0; a < plan->agg_count; a++) {
            YtValue* v = &dst[1 + a];
            v->id = (uint16_t)(1 + a);
            v->flags = 0;
            v->length = 0;
            if (plan->aggs[a]->func == YT_AGG_SUM1) {
                v->type = YT_VT_INT64;
                v->data.bits = sl->cnt;
            } else if (a == sum_idx && sl->nonnull) {
                v->type = sl->sum_dbl ? YT_VT_DOUBLE : YT_VT_INT64;
                v->data.bits = sl->sum_bits;
            } else {
                v->type = YT_VT_NULL;
                v->data.bits = 0;
            }
        }
        output->row_count++;
    }
    free(slots);
    return YT_OK;
}

ORACLE_EXPORT
int yto_decode_string_column(const YtColumn* col, int64_t row_count,
                             char* out_blob, int64_t blob_cap,
                             int64_t* out_end /* row_count+1 */, uint8_t* nulls)
{
    int64_t row = 0;
    int64_t used = 0;
    out_end[0] = 0;
    for (int s = 0; s < col->segment_count; s++) {
        const YtSegment* seg = &col->segments[s];
        int rc = decode_string_segment(seg, out_blob, blob_cap, &used,
                                       out_end + row, nulls + row);
        if (rc != 0) return rc == -2 ? YT_ERR_CAPACITY : YT_ERR_INVALID_CHUNK;
        row += seg->row_count;
    }
    return (row == row_count) ? YT_OK : YT_ERR_INVALID_CHUNK;
}

/* ------------------------------------------------------------------ */
/* versioned scan-format slice: restated reader.
 * Layouts: include/ytql_gpu.h citations (timestamp_writer.cpp DumpSegment,
 * column_writer_detail.cpp DumpVersionedData, integer_column_writer.cpp
 * DumpDirectValues). Read-at-timestamp rule restates
 * rowset_builder.cpp:1042-1166 (TRowAllocatorBase::DoAllocateRow,
 * produceAll = false): per row, timestamps DESC; lower = first ts <= T;
 * deleteTs = latest delete <= T (0 = none); visible writes are those <= T
 * and > deleteTs; the column value is the first value (values sorted by
 * ascending timestamp index) whose index lies in that range. */

static uint64_t vcum_at(const BitReader* diffs, uint32_t expected, int64_t i)
{
    if (i < 0) return 0;
    return (uint64_t)expected * (uint64_t)(i + 1)
         + (uint64_t)(int64_t)zigzag_decode32((uint32_t)bitreader_get(diffs, i));
}

ORACLE_EXPORT
int yto_versioned_read(const YtVersionedColumn* col, uint64_t timestamp,
                       uint64_t* out_bits, uint8_t* out_null,
                       uint8_t* out_visible, uint8_t* out_agg,
                       char* errbuf, size_t errlen)
{
    int64_t base_row = 0;
    for (int si = 0; si < col->ts_seg_count; si++) {
        const YtTimestampSeg* T = &col->ts_segs[si];
        const YtVersionedValueSeg* V = &col->val_segs[si];
        if (V->row_count != T->row_count) {
            set_err(errbuf, errlen, "versioned: segment row mismatch");
            return YT_ERR_INVALID_CHUNK;
        }
        const char* tp = (const char*)T->data;
        BitReader dict = bitreader_init(tp);
        tp += bitreader_byte_size(&dict);
        BitReader wids = bitreader_init(tp);
        tp += bitreader_byte_size(&wids);
        BitReader dids = bitreader_init(tp);
        tp += bitreader_byte_size(&dids);
        BitReader wdiffs = bitreader_init(tp);
        tp += bitreader_byte_size(&wdiffs);
        BitReader ddiffs = bitreader_init(tp);

        const char* vp = (const char*)V->data;
        const int sparse = (V->type & 2) != 0;
        const int is_str = V->type >= YT_VSEG_STR_DIRECT_DENSE;
        const int is_dbl = !is_str && V->type >= YT_VSEG_DOUBLE_DENSE;
        const int is_dict = !is_dbl && !is_str && (V->type & 1) != 0;
        const int is_sdict = is_str && (V->type & 1) != 0;
        BitReader vindex = bitreader_init(vp);    /* dense offsets | sparse row idx */
        vp += bitreader_byte_size(&vindex);
        BitReader tsids = bitreader_init(vp);
        vp += bitreader_byte_size(&tsids);
        const uint8_t* vaggbm = NULL;
        if (V->flags & YT_VSEG_F_AGGREGATE) {
            vaggbm = (const uint8_t*)vp;
            vp += ((tsids.size + 7) / 8 + 7) & ~(uint64_t)7;
        }
        BitReader vvals = {0};                    /* direct values | dictionary */
        BitReader vids = {0};                     /* dictionary ids */
        const double* ddata = NULL;
        const uint8_t* vnull = NULL;
        const char* sdata = NULL;                 /* string bytes / dict bytes */
        if (is_str) {
            /* string_column_writer.cpp DumpDirectValues/DumpDictionaryValues:
             * direct = [END offsets diff-from-expected][null bitmap][bytes];
             * dict = [ids 0=null][dict END offsets diff-from-expected][bytes] */
            if (is_sdict) {
                vids = bitreader_init(vp);
                vp += bitreader_byte_size(&vids);
                vvals = bitreader_init(vp);       /* dictionary offsets */
                vp += bitreader_byte_size(&vvals);
                sdata = vp;
            } else {
                vvals = bitreader_init(vp);       /* value offsets */
                vp += bitreader_byte_size(&vvals);
                vnull = (const uint8_t*)vp;
                vp += ((vvals.size + 7) / 8 + 7) & ~(uint64_t)7;
                sdata = vp;
            }
        } else if (is_dbl) {
            uint64_t cnt;
            memcpy(&cnt, vp, 8);
            ddata = (const double*)(vp + 8);
            vnull = (const uint8_t*)(vp + 8 + cnt * 8);
        } else if (is_dict) {
            vvals = bitreader_init(vp);
            vp += bitreader_byte_size(&vvals);
            vids = bitreader_init(vp);
        } else {
            vvals = bitreader_init(vp);
            vp += bitreader_byte_size(&vvals);
            vnull = (const uint8_t*)vp;
        }

        for (int64_t r = 0; r < T->row_count; r++) {
            int64_t g = base_row + r;
            out_visible[g] = 0;
            out_null[g] = 1;
            out_bits[g] = 0;
            if (out_agg) out_agg[g] = 0;
            uint64_t wb = vcum_at(&wdiffs, T->expected_writes_per_row, r - 1);
            uint64_t we = vcum_at(&wdiffs, T->expected_writes_per_row, r);
            uint64_t db = vcum_at(&ddiffs, T->expected_deletes_per_row, r - 1);
            uint64_t de = vcum_at(&ddiffs, T->expected_deletes_per_row, r);

            /* latest delete <= T (lists are DESC: scan past those > T) */
            uint64_t delete_ts = 0;
            for (uint64_t i = db; i < de; i++) {
                uint64_t ts = T->base_timestamp + bitreader_get(&dict, bitreader_get(&dids, i));
                if (ts <= timestamp) { delete_ts = ts; break; }
            }
            /* lower/upper write indexes within the row (0-based in-row) */
            int64_t wcount = (int64_t)(we - wb);
            int64_t lower = wcount, upper = wcount;
            for (int64_t i = 0; i < wcount; i++) {
                uint64_t ts = T->base_timestamp + bitreader_get(&dict, bitreader_get(&wids, wb + i));
                if (lower == wcount && ts <= timestamp) lower = i;
                if (ts <= delete_ts) { upper = i; break; }
            }
            if (upper < lower) upper = lower;
            if (lower >= upper) continue;     /* no visible write: row absent */
            out_visible[g] = 1;

            /* first value with in-row timestamp index in [lower, upper) */
            uint64_t vb, ve;
            if (sparse) {
                /* row indexes ascending: binary search the run for r */
                uint64_t lo2 = 0, hi2 = vindex.size;
                while (lo2 < hi2) {
                    uint64_t mid = (lo2 + hi2) >> 1;
                    if ((int64_t)bitreader_get(&vindex, mid) < r) lo2 = mid + 1;
                    else hi2 = mid;
                }
                vb = lo2;
                hi2 = vindex.size;
                while (lo2 < hi2) {
                    uint64_t mid = (lo2 + hi2) >> 1;
                    if ((int64_t)bitreader_get(&vindex, mid) <= r) lo2 = mid + 1;
                    else hi2 = mid;
                }
                ve = lo2;
            } else {
                vb = vcum_at(&vindex, V->expected_values_per_row, r - 1);
                ve = vcum_at(&vindex, V->expected_values_per_row, r);
            }
            for (uint64_t j = vb; j < ve; j++) {
                uint64_t ti = bitreader_get(&tsids, j);
                if ((int64_t)ti < lower) continue;
                if ((int64_t)ti >= upper) break;
                int nul;
                uint64_t bits = 0;
                if (is_str) {
                    /* bits = (offset within V->data) << 24 | length */
                    if (is_sdict) {
                        uint64_t id = bitreader_get(&vids, j);
                        nul = (id == 0);
                        if (!nul) {
                            uint64_t e2 = vcum_at(&vvals, (uint32_t)V->base_value,
                                                  (int64_t)id - 1);
                            uint64_t b2 = vcum_at(&vvals, (uint32_t)V->base_value,
                                                  (int64_t)id - 2);
                            bits = ((uint64_t)(sdata - (const char*)V->data + b2) << 24)
                                 | (e2 - b2);
                        }
                    } else {
                        nul = (vnull[j / 8] >> (j % 8)) & 1;
                        if (!nul) {
                            uint64_t e2 = vcum_at(&vvals, (uint32_t)V->base_value,
                                                  (int64_t)j);
                            uint64_t b2 = vcum_at(&vvals, (uint32_t)V->base_value,
                                                  (int64_t)j - 1);
                            bits = ((uint64_t)(sdata - (const char*)V->data + b2) << 24)
                                 | (e2 - b2);
                        }
                    }
                } else if (is_dict) {
                    uint64_t id = bitreader_get(&vids, j);
                    nul = (id == 0);
                    if (!nul)
                        bits = (uint64_t)zigzag_decode64(
                            V->base_value + bitreader_get(&vvals, id - 1));
                } else {
                    nul = (vnull[j / 8] >> (j % 8)) & 1;
                    if (!nul) {
                        if (is_dbl) memcpy(&bits, &ddata[j], 8);
                        else bits = (uint64_t)zigzag_decode64(
                            V->base_value + bitreader_get(&vvals, j));
                    }
                }
                if (!nul) {
                    out_null[g] = 0;
                    out_bits[g] = bits;
                }
                /* aggregate flag of the chosen value, null or not */
                if (out_agg && vaggbm)
                    out_agg[g] = (vaggbm[j / 8] >> (j % 8)) & 1;
                break;
            }
        }
        base_row += T->row_count;
    }
    return YT_OK;
}
