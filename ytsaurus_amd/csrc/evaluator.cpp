/* evaluator.cpp — host side of the MI355X query executor: the C-ABI
 * implementation of the IEvaluator::Run seam (reference
 * engine/evaluator.cpp:51-105), plan compilation (AOT replacement of the
 * LLVM codegen in folding_profiler.cpp / cg_fragment_compiler.cpp), device
 * orchestration, limits and statistics (TExecutionContext /
 * TQueryStatistics, engine_api/evaluation_helpers.h:251-280).
 *
 * Product code. No CPU fallback: without a HIP device every execute entry
 * returns YT_ERR_NO_GPU / YT_ERR_HIP.
 */
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <string.h>
#include <stdio.h>
#include <stdlib.h>
#include <vector>
#include <algorithm>
#include <mutex>
#include <chrono>
#include <thread>
#include <atomic>

#include "common.h"
#include "../../include/ytql_gpu.h"

using namespace ytql;

/* launch wrappers from kernels.hip */
extern "C" {
hipError_t ytql_launch_parse_segments(const DevSeg*, int, SegEx*, unsigned*, unsigned*, unsigned long long*, unsigned long long*, hipStream_t);
hipError_t ytql_launch_scan_nullflags(const DevSeg*, const SegEx*, int, unsigned*, hipStream_t);
hipError_t ytql_launch_scan_zzrange(const DevSeg*, const SegEx*, int, int, int,
                                    unsigned long long*, hipStream_t);
hipError_t ytql_launch_scan_partition(const PartParams*, const DevSeg*, const SegEx*,
                                      const FastCol*, TableHdr*, unsigned long long*,
                                      void*, unsigned long long*, uint64_t*,
                                      size_t, int, hipStream_t);
hipError_t ytql_launch_bucket_agg(const void*, const unsigned long long*, int64_t,
                                  const uint64_t*, const unsigned long long*, int64_t,
                                  OutGroup*, unsigned long long*, int64_t,
                                  TableHdr*, int, int, int, int,
                                  uint64_t, uint64_t, int, int, hipStream_t);
hipError_t ytql_launch_bucket_agg_direct(const void*, const unsigned long long*, int64_t,
                                         const uint64_t*, const unsigned long long*, int64_t,
                                         OutGroup*, unsigned long long*, int64_t,
                                         TableHdr*, int, int, int, int,
                                         uint64_t, uint64_t, int, int, hipStream_t);
hipError_t ytql_launch_topk_hist(const DevPlan*, const DevSeg*, const SegEx*,
                                 const int32_t*, const int32_t*, int64_t,
                                 const JoinDev*, const JoinDev*,
                                 const TopkPass*, unsigned long long*,
                                 unsigned long long*, unsigned*, hipStream_t);
hipError_t ytql_launch_topk_gather(const DevPlan*, const DevSeg*, const SegEx*,
                                   const int32_t*, const int32_t*, int64_t,
                                   const JoinDev*, const JoinDev*,
                                   const TopkGather*,
                                   int64_t*, unsigned long long*,
                                   int64_t*, unsigned long long*,
                                   int64_t*, unsigned long long*,
                                   unsigned*, hipStream_t);
hipError_t ytql_launch_topk_materialize(const DevPlan*, const DevSeg*, const SegEx*,
                                        const int32_t*, const int32_t*,
                                        const JoinDev*, const JoinDev*,
                                        const int64_t*, int64_t, DevOutVal*,
                                        unsigned*, hipStream_t);
hipError_t ytql_launch_topk_hist_fast(const DevSeg*, const SegEx*, int, int,
                                      int64_t, int, int, const TopkPass*,
                                      unsigned long long*, unsigned long long*,
                                      hipStream_t);
hipError_t ytql_launch_topk_gather_fast(const DevSeg*, const SegEx*, int, int,
                                        int64_t, int, int, const TopkGather*,
                                        int64_t*, unsigned long long*,
                                        int64_t*, unsigned long long*,
                                        int64_t*, unsigned long long*,
                                        hipStream_t);
hipError_t ytql_launch_versioned_read(const VSegDev*, int, int64_t, uint64_t,
                                      uint64_t*, uint8_t*, uint8_t*, uint8_t*,
                                      hipStream_t);
hipError_t ytql_launch_vis_count(const uint8_t*, int64_t,
                                 unsigned long long*, int, hipStream_t);
hipError_t ytql_launch_unvcol_to_arrays(const DevSeg*, const SegEx*, int, int,
                                        int64_t, uint64_t*, uint8_t*,
                                        unsigned*, hipStream_t);
hipError_t ytql_launch_vis_scatter(const uint8_t*, const uint8_t*,
                                   const uint64_t*, int64_t,
                                   const unsigned long long*, uint64_t,
                                   const int64_t*, char*, int, int, hipStream_t);
hipError_t ytql_launch_join_build(const JoinDev*, int64_t, uint64_t*,
                                  long long*, unsigned long long*, unsigned*,
                                  hipStream_t);
hipError_t ytql_launch_join_chain(const JoinDev*, int64_t, unsigned*,
                                   hipStream_t);
hipError_t ytql_launch_scan_project(const DevPlan*, const DevSeg*, const SegEx*,
                                    const int32_t*, const int32_t*, int64_t,
                                    int64_t, const JoinDev*, const JoinDev*,
                                    DevOutVal*, uint8_t*, unsigned*, hipStream_t);
hipError_t ytql_launch_scan_generic(const DevPlan*, const DevSeg*, const SegEx*,
                                    const int32_t*, const int32_t*, int64_t,
                                    const JoinDev*, const JoinDev*,
                                    TableHdr*, unsigned long long*, unsigned*, hipStream_t);
hipError_t ytql_launch_scan_fast(const FastParams*, const DevSeg*, const SegEx*,
                                 const FastCol*, TableHdr*, unsigned long long*,
                                 unsigned long long*, size_t, int, hipStream_t);
hipError_t ytql_launch_compact(TableHdr*, TableHdr*, const unsigned long long*, int,
                               OutGroup*, unsigned long long*, uint64_t, hipStream_t);
hipError_t ytql_launch_part_count(const OutGroup*, int64_t, int, int,
                                  unsigned long long*, hipStream_t);
hipError_t ytql_launch_part_scatter(const OutGroup*, int64_t, int, int, int,
                                    int, unsigned long long*, YtStateRow*,
                                    hipStream_t);
hipError_t ytql_launch_merge_states(const YtStateRow*, int64_t, int, int,
                                    int, TableHdr*, unsigned long long*,
                                    hipStream_t);
hipError_t ytql_launch_strgrp_accum(const StrGroupParams*, const DevSeg*, const SegEx*,
                                    const int64_t*, unsigned long long*, TableHdr*,
                                    hipStream_t);
hipError_t ytql_launch_strgrp_hash(const DevSeg*, const SegEx*, int, int,
                                   const int64_t*, uint64_t*, uint64_t*,
                                   ulonglong2*, int64_t, hipStream_t);
hipError_t ytql_launch_strgrp_merge(const DevSeg*, const SegEx*, int, int,
                                    const int64_t*, const unsigned long long*,
                                    const uint64_t*, const uint64_t*,
                                    const ulonglong2*, StrSlot*, uint64_t, int,
                                    int, TableHdr*, int64_t, hipStream_t);
hipError_t ytql_launch_strgrp_compact(const DevSeg*, const SegEx*, int,
                                      const StrSlot*, uint64_t, OutStrGroup*,
                                      unsigned long long*, char*, unsigned long long*,
                                      uint64_t, TableHdr*, hipStream_t);
hipError_t ytql_launch_strst_count(const OutStrGroup*, int64_t, const char*,
                                   int, unsigned long long*,
                                   unsigned long long*, hipStream_t);
hipError_t ytql_launch_strst_scatter(const OutStrGroup*, int64_t, const char*,
                                     int, int, unsigned long long*,
                                     unsigned long long*, YtStateRow*, char*,
                                     const unsigned long long*, hipStream_t);
hipError_t ytql_launch_strst_hash(const YtStateRow*, int64_t, const int64_t*,
                                  const unsigned long long*, int, const char*,
                                  uint64_t*, uint64_t*, ulonglong2*,
                                  uint64_t*, hipStream_t);
hipError_t ytql_launch_strst_merge(const YtStateRow*, int64_t, const char*,
                                   const uint64_t*, const uint64_t*,
                                   const ulonglong2*, const uint64_t*, int,
                                   StrSlot*, uint64_t, TableHdr*, hipStream_t);
hipError_t ytql_launch_strst_compact(const StrSlot*, uint64_t,
                                     const uint64_t*, const char*,
                                     OutStrState*, unsigned long long*,
                                     char*, unsigned long long*, uint64_t,
                                     TableHdr*, hipStream_t);
}

/* must match kernels.hip mix64 and the oracle's splitmix64 */
static inline uint64_t splitmix64_host(uint64_t x)
{
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

/* partition-output request for run_string_group (string-keyed bottom
 * query): when set, the compacted groups are hash-partitioned into
 * state rows + pool slices instead of being emitted as a rowset. */
struct StrPartialOut {
    int32_t partition_count;
    YtStateRow* states_device;
    int64_t capacity_rows;
    char* pool_device;
    int64_t pool_capacity;
    int64_t* part_counts;        /* out */
    int64_t* part_pool_bytes;    /* out */
};

/* ------------------------------------------------------------------ */
/* device/pinned buffer pool: query executions reuse large allocations
 * across calls (the reference's evaluator similarly reuses codegen and
 * memory-chunk pools per query class, evaluator.cpp:256, TExpressionContext
 * chunk providers). Freed via yt_gpu_pool_trim(). */

namespace {

struct Pool {
    struct Blk { void* p; size_t sz; bool used; bool host; };
    std::mutex m;
    std::vector<Blk> blks;

    hipError_t get(size_t sz, bool host, void** out)
    {
        std::lock_guard<std::mutex> g(m);
        Blk* best = nullptr;
        for (auto& b : blks) {
            if (!b.used && b.host == host && b.sz >= sz &&
                (!best || b.sz < best->sz)) {
                best = &b;
            }
        }
        if (best && best->sz <= sz * 2 + (64 << 20)) {
            best->used = true;
            *out = best->p;
            return hipSuccess;
        }
        void* p = nullptr;
        hipError_t e = host ? hipHostMalloc(&p, sz) : hipMalloc(&p, sz);
        if (e != hipSuccess) {
            /* retry after trimming the pool */
            for (auto& b : blks) {
                if (!b.used) {
                    if (b.host) { (void)hipHostFree(b.p); } else { (void)hipFree(b.p); }
                    b.p = nullptr;
                    b.sz = 0;
                }
            }
            blks.erase(std::remove_if(blks.begin(), blks.end(),
                                      [](const Blk& b) { return !b.p; }),
                       blks.end());
            e = host ? hipHostMalloc(&p, sz) : hipMalloc(&p, sz);
            if (e != hipSuccess) return e;
        }
        blks.push_back({p, sz, true, host});
        *out = p;
        return hipSuccess;
    }

    void put(void* p)
    {
        if (!p) return;
        std::lock_guard<std::mutex> g(m);
        for (auto& b : blks) {
            if (b.p == p) { b.used = false; return; }
        }
    }

    void trim()
    {
        std::lock_guard<std::mutex> g(m);
        for (auto& b : blks) {
            if (!b.used) {
                if (b.host) { (void)hipHostFree(b.p); } else { (void)hipFree(b.p); }
                b.p = nullptr;
            }
        }
        blks.erase(std::remove_if(blks.begin(), blks.end(),
                                  [](const Blk& b) { return !b.p; }),
                   blks.end());
    }
};

Pool g_pool;

template <class T>
hipError_t pool_alloc(T** p, size_t sz)
{
    return g_pool.get(sz, false, (void**)p);
}

template <class T>
hipError_t pool_alloc_host(T** p, size_t sz)
{
    return g_pool.get(sz, true, (void**)p);
}

double now_ms()
{
    return std::chrono::duration<double, std::milli>(
        std::chrono::steady_clock::now().time_since_epoch()).count();
}

} /* namespace */

extern "C" void yt_gpu_pool_trim(void)
{
    g_pool.trim();
}

static void set_err(char* errbuf, size_t errlen, const char* msg)
{
    if (errbuf && errlen) snprintf(errbuf, errlen, "%s", msg);
}

#define HIP_CHECK(call)                                                     \
    do {                                                                    \
        hipError_t e_ = (call);                                             \
        if (e_ != hipSuccess) {                                             \
            if (errbuf) snprintf(errbuf, errlen, "HIP error %s at %s:%d",   \
                                 hipGetErrorString(e_), __FILE__, __LINE__);\
            rc = YT_ERR_HIP;                                                \
            goto fail;                                                      \
        }                                                                   \
    } while (0)

extern "C" int yt_gpu_available(char* errbuf, size_t errlen)
{
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess || n == 0) {
        set_err(errbuf, errlen, "no HIP device");
        return YT_ERR_NO_GPU;
    }
    return YT_OK;
}

/* ------------------------------------------------------------------ */
/* plan compilation: YtExpr tree → postfix DevPlan program              */

static int compile_expr(const YtExpr* e, DevPlan* p, int* len, char* errbuf, size_t errlen)
{
    if (!e) { set_err(errbuf, errlen, "null expr"); return YT_ERR_INVALID_PLAN; }
    switch (e->op) {
    case YT_EX_COLUMN:
    case YT_EX_LIT_I64:
    case YT_EX_LIT_NULL:
    case YT_EX_LIT_DOUBLE:
        break;
    case YT_EX_NOT: {
        int rc = compile_expr(e->a, p, len, errbuf, errlen);
        if (rc) return rc;
        break;
    }
    default: {
        int rc = compile_expr(e->a, p, len, errbuf, errlen);
        if (rc) return rc;
        rc = compile_expr(e->b, p, len, errbuf, errlen);
        if (rc) return rc;
        break;
    }
    }
    if (p->prog_len >= kMaxProg) { set_err(errbuf, errlen, "expression too long"); return YT_ERR_INVALID_PLAN; }
    PInst& in = p->prog[p->prog_len++];
    in.op = e->op;     /* YT_EX_* values coincide with P_* by construction */
    in.col = e->col;
    in.bits = 0;
    if (e->op == YT_EX_LIT_I64) in.bits = (uint64_t)e->lit_i64;
    if (e->op == YT_EX_LIT_DOUBLE) memcpy(&in.bits, &e->lit_dbl, 8);
    if (e->op == YT_EX_COLUMN && (e->col < 0 || e->col >= p->ncols)) {
        set_err(errbuf, errlen, "column index out of range");
        return YT_ERR_INVALID_PLAN;
    }
    (*len)++;
    return YT_OK;
}

/* static result type of an expression (for output formatting) */
static uint8_t expr_static_type(const YtExpr* e, const uint8_t* col_types)
{
    switch (e->op) {
    case YT_EX_COLUMN: return col_types[e->col];
    case YT_EX_LIT_I64: return YT_VT_INT64;
    case YT_EX_LIT_DOUBLE: return YT_VT_DOUBLE;
    case YT_EX_LIT_NULL: return YT_VT_NULL;
    case YT_EX_NOT: return YT_VT_BOOLEAN;
    default:
        if (e->op >= YT_EX_EQ) return YT_VT_BOOLEAN;   /* cmp / and / or */
        return expr_static_type(e->a, col_types);      /* arithmetic */
    }
}

static bool expr_is_col(const YtExpr* e, int* col);

static int build_devplan(const YtPlan* plan, const YtChunk* chunk, DevPlan* p,
                         char* errbuf, size_t errlen)
{
    memset(p, 0, sizeof(*p));
    p->ncols = chunk->column_count;
    for (const YtJoin* J = plan->join; J; J = J->next)
        p->ncols += J->foreign_value_count;
    if (p->ncols > kMaxCols) { set_err(errbuf, errlen, "too many columns"); return YT_ERR_UNSUPPORTED; }
    for (int c = 0; c < chunk->column_count; c++) p->col_types[c] = (uint8_t)chunk->columns[c].value_type;
    {
        int at = chunk->column_count;
        for (const YtJoin* J = plan->join; J; J = J->next)
            for (int j = 0; j < J->foreign_value_count; j++)
                p->col_types[at++] =
                    (uint8_t)J->foreign->columns[J->foreign_value_cols[j]].value_type;
    }
    for (int c = 0; c < chunk->column_count; c++) {
        const YtColumn& col = chunk->columns[c];
        int shift = 0;
        if (col.segment_count > 0) {
            int32_t r0 = col.segments[0].row_count;
            if (r0 > 0 && (r0 & (r0 - 1)) == 0) {
                int ok = 1;
                for (int i = 0; i + 1 < col.segment_count; i++)
                    if (col.segments[i].row_count != r0) { ok = 0; break; }
                if (ok && col.segments[col.segment_count - 1].row_count <= r0) {
                    while ((1 << (shift + 1)) <= r0) shift++;
                    if ((1 << shift) != r0) shift = 0;
                }
            }
        }
        p->col_uniform_shift[c] = shift;
    }

    if (plan->key_count > 1) {
        /* composite packed key: plain int64/uint64/boolean key columns,
         * widths resolved after segment parse (see pack_group_key) */
        if (plan->key_count > kMaxPackKeys) {
            set_err(errbuf, errlen, "GPU path: at most 4 group keys this round");
            return YT_ERR_UNSUPPORTED;
        }
        p->kp_count = plan->key_count;
        for (int i = 0; i < plan->key_count; i++) {
            int c;
            if (!expr_is_col(plan->keys[i], &c) || c >= p->ncols) {
                set_err(errbuf, errlen,
                        "multi-key GROUP BY: plain key columns this round");
                return YT_ERR_UNSUPPORTED;
            }
            if (c >= chunk->column_count) {
                set_err(errbuf, errlen,
                        "multi-key GROUP BY over joined columns: not this round");
                return YT_ERR_UNSUPPORTED;
            }
            uint8_t vt = p->col_types[c];
            if (vt != YT_VT_INT64 && vt != YT_VT_UINT64 && vt != YT_VT_BOOLEAN) {
                set_err(errbuf, errlen,
                        "multi-key GROUP BY: int64/uint64/boolean keys this round");
                return YT_ERR_UNSUPPORTED;
            }
            p->kp_col[i] = c;
            p->kp_signed[i] = vt == YT_VT_INT64;
        }
    }
    if (plan->agg_count == 0 && plan->key_count == 0) {
        if (plan->project_count < 1 || plan->project_count > kMaxProj) {
            set_err(errbuf, errlen, "scan mode needs 1..8 projections");
            return YT_ERR_UNSUPPORTED;
        }
    } else if (plan->agg_count > 0)
    if (plan->agg_count < 1 || plan->agg_count > kMaxAggs) { set_err(errbuf, errlen, "need 1..4 aggregates"); return YT_ERR_UNSUPPORTED; }
    for (int a = 0; a < plan->agg_count; a++) {
        int f = plan->aggs[a]->func;
        if (f < YT_AGG_SUM || f > YT_AGG_AVG) {
            set_err(errbuf, errlen, "unsupported aggregate");
            return YT_ERR_UNSUPPORTED;
        }
    }

    int rc;
    if (plan->filter) {
        p->filter_off = p->prog_len;
        rc = compile_expr(plan->filter, p, &p->filter_len, errbuf, errlen);
        if (rc) return rc;
        p->filter_len = p->prog_len - p->filter_off;
    }
    if (plan->key_count == 1) {
        p->key_off = p->prog_len;
        rc = compile_expr(plan->keys[0], p, &p->key_len, errbuf, errlen);
        if (rc) return rc;
        p->key_len = p->prog_len - p->key_off;
    }
    p->agg_count = plan->agg_count;
    for (int a = 0; a < plan->agg_count; a++) {
        p->agg_func[a] = plan->aggs[a]->func;
        if (plan->aggs[a]->func < YT_AGG_SUM ||
            plan->aggs[a]->func > YT_AGG_AVG) {
            set_err(errbuf, errlen, "unknown aggregate function");
            return YT_ERR_UNSUPPORTED;
        }
        if (plan->aggs[a]->func == YT_AGG_FIRST ||
            plan->aggs[a]->func == YT_AGG_AVG) {
            /* no packed state format for these: strings can't live in one
             * slot word (first), and avg-of-avgs is wrong (totals fold) */
            uint8_t at = expr_static_type(plan->aggs[a]->arg, p->col_types);
            if (plan->aggs[a]->func == YT_AGG_AVG &&
                at != YT_VT_INT64 && at != YT_VT_UINT64 && at != YT_VT_DOUBLE) {
                set_err(errbuf, errlen, "avg: int64/uint64/double argument only");
                return YT_ERR_UNSUPPORTED;
            }
            if (plan->aggs[a]->func == YT_AGG_FIRST && at == YT_VT_STRING) {
                set_err(errbuf, errlen,
                        "first(): string argument is oracle-only this round");
                return YT_ERR_UNSUPPORTED;
            }
            if (plan->aggs[a]->func == YT_AGG_AVG && plan->with_totals) {
                set_err(errbuf, errlen, "avg: WITH TOTALS not this round");
                return YT_ERR_UNSUPPORTED;
            }
        }
        if (plan->aggs[a]->func != YT_AGG_SUM1) {
            p->agg_off[a] = p->prog_len;
            rc = compile_expr(plan->aggs[a]->arg, p, &p->agg_len[a], errbuf, errlen);
            if (rc) return rc;
            p->agg_len[a] = p->prog_len - p->agg_off[a];
        }
    }
    if (plan->agg_count == 0) {
        p->proj_count = plan->project_count;
        for (int pj = 0; pj < plan->project_count; pj++) {
            p->proj_off[pj] = p->prog_len;
            rc = compile_expr(plan->projects[pj], p, &p->proj_len[pj], errbuf, errlen);
            if (rc) return rc;
            p->proj_len[pj] = p->prog_len - p->proj_off[pj];
        }
    }
    return YT_OK;
}

/* fast-shape analysis: filter = none | cmp(col, lit) | AND of two bounds;
 * key = direct int64 column | none; sum args = direct int64 columns */
static bool expr_is_col(const YtExpr* e, int* col)
{
    if (e && e->op == YT_EX_COLUMN) { *col = e->col; return true; }
    return false;
}

static bool match_bound(const YtExpr* e, const uint8_t* col_types,
                        int* col, int64_t* lo, int64_t* hi)
{
    /* col cmp lit / lit cmp col, int64 only */
    int c;
    int64_t k;
    int op = e->op;
    if (op < YT_EX_EQ || op > YT_EX_GE || op == YT_EX_NE) return false;
    if (expr_is_col(e->a, &c) && e->b && e->b->op == YT_EX_LIT_I64) {
        k = e->b->lit_i64;
    } else if (expr_is_col(e->b, &c) && e->a && e->a->op == YT_EX_LIT_I64) {
        k = e->a->lit_i64;
        /* mirror: lit cmp col → col cmp' lit */
        switch (op) {
        case YT_EX_LT: op = YT_EX_GT; break;
        case YT_EX_LE: op = YT_EX_GE; break;
        case YT_EX_GT: op = YT_EX_LT; break;
        case YT_EX_GE: op = YT_EX_LE; break;
        default: break;
        }
    } else {
        return false;
    }
    if (col_types[c] != YT_VT_INT64) return false;
    *col = c;
    *lo = INT64_MIN; *hi = INT64_MAX;
    switch (op) {
    case YT_EX_EQ: *lo = *hi = k; break;
    case YT_EX_LT: if (k == INT64_MIN) return false; *hi = k - 1; break;
    case YT_EX_LE: *hi = k; break;
    case YT_EX_GT: if (k == INT64_MAX) return false; *lo = k + 1; break;
    case YT_EX_GE: *lo = k; break;
    default: return false;
    }
    return true;
}

static void analyze_fast(const YtPlan* plan, const YtChunk* chunk, FastShape* fs)
{
    memset(fs, 0, sizeof(*fs));
    fs->filter_col = -1;
    fs->key_col = -1;
    if (plan->join) return;           /* joined plans run the generic path */

    const uint8_t* ct = nullptr;
    static uint8_t types[kMaxCols];
    for (int c = 0; c < chunk->column_count && c < kMaxCols; c++)
        types[c] = (uint8_t)chunk->columns[c].value_type;
    ct = types;

    /* all segments of all columns must be DirectDense int64, uniform interior
     * row counts */
    int32_t seg_cnt0 = chunk->columns[0].segment_count;
    int32_t rows0 = seg_cnt0 ? chunk->columns[0].segments[0].row_count : 0;
    for (int c = 0; c < chunk->column_count; c++) {
        const YtColumn& col = chunk->columns[c];
        if (col.value_type != YT_VT_INT64) return;
        if (col.segment_count != seg_cnt0) return;
        for (int s = 0; s < col.segment_count; s++) {
            if (col.segments[s].type != YT_SEG_DIRECT_DENSE) return;
            if (s < col.segment_count - 1 && col.segments[s].row_count != rows0) return;
            if (col.segments[s].row_count > rows0) return;
        }
    }
    if (seg_cnt0 == 0) return;

    if (plan->filter) {
        int64_t lo, hi;
        int col;
        if (match_bound(plan->filter, ct, &col, &lo, &hi)) {
            fs->filter_col = col; fs->filter_lo = lo; fs->filter_hi = hi;
        } else if (plan->filter->op == YT_EX_AND) {
            int c1, c2;
            int64_t lo1, hi1, lo2, hi2;
            if (!match_bound(plan->filter->a, ct, &c1, &lo1, &hi1)) return;
            if (!match_bound(plan->filter->b, ct, &c2, &lo2, &hi2)) return;
            if (c1 != c2) return;
            fs->filter_col = c1;
            fs->filter_lo = std::max(lo1, lo2);
            fs->filter_hi = std::min(hi1, hi2);
        } else {
            return;
        }
    }
    if (plan->key_count == 1) {
        int col;
        if (!expr_is_col(plan->keys[0], &col)) return;
        if (ct[col] != YT_VT_INT64) return;
        fs->key_col = col;
    } else if (plan->key_count > 1) {
        return;
    }
    fs->nsum = 0;
    for (int a = 0; a < plan->agg_count; a++) {
        if (plan->aggs[a]->func == YT_AGG_SUM1) { fs->have_sum1 = 1; continue; }
        if (plan->aggs[a]->func != YT_AGG_SUM) return;
        int col;
        if (!expr_is_col(plan->aggs[a]->arg, &col)) return;
        if (ct[col] != YT_VT_INT64) return;
        if (fs->nsum >= kMaxAggs) return;
        fs->sum_col[fs->nsum] = col;
        fs->sum_slot[fs->nsum] = a;
        fs->nsum++;
    }
    if (fs->nsum == 0 && fs->filter_col < 0 && fs->key_col < 0) {
        return;   /* pure row count with no staged column — generic path */
    }
    fs->valid = 1;
}

/* ------------------------------------------------------------------ */
/* host-side projection evaluation over a finalized group row           */
/* (mirrors MakeCodegenProjectOp over [keys..., aggregates...])        */

struct HVal { uint8_t type; uint64_t bits; };

static int heval(const YtExpr* e, const HVal* row, int nrow, HVal* out)
{
    switch (e->op) {
    case YT_EX_COLUMN:
        if (e->col < 0 || e->col >= nrow) return YT_ERR_INVALID_PLAN;
        *out = row[e->col];
        return YT_OK;
    case YT_EX_LIT_I64: out->type = YT_VT_INT64; out->bits = (uint64_t)e->lit_i64; return YT_OK;
    case YT_EX_LIT_DOUBLE: out->type = YT_VT_DOUBLE; memcpy(&out->bits, &e->lit_dbl, 8); return YT_OK;
    case YT_EX_LIT_NULL: out->type = YT_VT_NULL; out->bits = 0; return YT_OK;
    case YT_EX_NOT: {
        HVal a;
        int rc = heval(e->a, row, nrow, &a);
        if (rc) return rc;
        if (a.type == YT_VT_NULL) { *out = a; return YT_OK; }
        out->type = YT_VT_BOOLEAN; out->bits = !a.bits;
        return YT_OK;
    }
    default: break;
    }
    HVal a, b;
    int rc = heval(e->a, row, nrow, &a);
    if (rc) return rc;
    rc = heval(e->b, row, nrow, &b);
    if (rc) return rc;
    out->type = YT_VT_NULL; out->bits = 0;
    if (e->op >= YT_EX_ADD && e->op <= YT_EX_MOD) {
        if (a.type == YT_VT_NULL || b.type == YT_VT_NULL) return YT_OK;
        if (a.type == YT_VT_DOUBLE) {
            double x, y, z = 0;
            memcpy(&x, &a.bits, 8);
            memcpy(&y, &b.bits, 8);
            switch (e->op) {
            case YT_EX_ADD: z = x + y; break;
            case YT_EX_SUB: z = x - y; break;
            case YT_EX_MUL: z = x * y; break;
            case YT_EX_DIV: z = x / y; break;
            default: return YT_ERR_UNSUPPORTED;
            }
            out->type = YT_VT_DOUBLE;
            memcpy(&out->bits, &z, 8);
            return YT_OK;
        }
        uint64_t x = a.bits, y = b.bits, z = 0;
        int sgn = (a.type == YT_VT_INT64);
        switch (e->op) {
        case YT_EX_ADD: z = x + y; break;
        case YT_EX_SUB: z = x - y; break;
        case YT_EX_MUL: z = x * y; break;
        case YT_EX_DIV:
        case YT_EX_MOD:
            if (y == 0) return YT_ERR_DIV_ZERO;
            if (sgn) {
                int64_t sx = (int64_t)x, sy = (int64_t)y;
                if (sx == INT64_MIN && sy == -1) z = (e->op == YT_EX_DIV) ? (uint64_t)INT64_MIN : 0;
                else z = (uint64_t)((e->op == YT_EX_DIV) ? sx / sy : sx % sy);
            } else {
                z = (e->op == YT_EX_DIV) ? x / y : x % y;
            }
            break;
        }
        out->type = a.type;
        out->bits = z;
        return YT_OK;
    }
    if (e->op >= YT_EX_EQ && e->op <= YT_EX_GE) {
        int lt, eq, force_true = 0;
        if (a.type == YT_VT_NULL || b.type == YT_VT_NULL) {
            unsigned ln = (a.type == YT_VT_NULL), rn = (b.type == YT_VT_NULL);
            lt = rn < ln; eq = ln == rn;
        } else if (a.type == YT_VT_DOUBLE) {
            double x, y;
            memcpy(&x, &a.bits, 8);
            memcpy(&y, &b.bits, 8);
            int unordered = (x != x) || (y != y);
            lt = unordered || (x < y);
            eq = unordered || (x == y);
            force_true = unordered;
        } else if (a.type == YT_VT_INT64) {
            int64_t x = (int64_t)a.bits, y = (int64_t)b.bits;
            lt = x < y; eq = x == y;
        } else {
            lt = a.bits < b.bits; eq = a.bits == b.bits;
        }
        int r = 1;
        if (!force_true) {
            switch (e->op) {
            case YT_EX_EQ: r = eq; break;
            case YT_EX_NE: r = !eq; break;
            case YT_EX_LT: r = lt; break;
            case YT_EX_LE: r = lt || eq; break;
            case YT_EX_GT: r = !(lt || eq); break;
            case YT_EX_GE: r = !lt; break;
            }
        }
        out->type = YT_VT_BOOLEAN;
        out->bits = (uint64_t)r;
        return YT_OK;
    }
    if (e->op == YT_EX_AND || e->op == YT_EX_OR) {
        int an = (a.type == YT_VT_NULL), bn = (b.type == YT_VT_NULL);
        int av = an ? 0 : (a.bits != 0), bv = bn ? 0 : (b.bits != 0);
        if (e->op == YT_EX_AND) {
            if ((!an && !av) || (!bn && !bv)) { out->type = YT_VT_BOOLEAN; out->bits = 0; }
            else if (!an && !bn) { out->type = YT_VT_BOOLEAN; out->bits = 1; }
        } else {
            if ((!an && av) || (!bn && bv)) { out->type = YT_VT_BOOLEAN; out->bits = 1; }
            else if (!an && !bn) { out->type = YT_VT_BOOLEAN; out->bits = 0; }
        }
        return YT_OK;
    }
    return YT_ERR_UNSUPPORTED;
}

/* ------------------------------------------------------------------ */
/* device chunk setup                                                  */

struct DeviceRun {
    std::vector<DevSeg> h_segs;
    std::vector<int32_t> h_off, h_cnt;
    DevSeg* d_segs = nullptr;
    SegEx* d_segex = nullptr;
    int32_t* d_off = nullptr;
    int32_t* d_cnt = nullptr;
    unsigned* d_maxw = nullptr;
    unsigned* d_err = nullptr;
    TableHdr* d_th = nullptr;
    unsigned long long* d_slots = nullptr;
    OutGroup* d_groups = nullptr;
    unsigned long long* d_counter = nullptr;
    unsigned long long* d_gaccum = nullptr;
    FastCol* d_fastcols = nullptr;
    unsigned* d_colnull = nullptr;
    unsigned long long* d_zzrange = nullptr;   /* [0..kMaxCols) min, [kMaxCols..) max */
    unsigned long long* d_cursors = nullptr;
    void* d_recs = nullptr;
    unsigned long long* d_ncursors = nullptr;
    uint64_t* d_nrecs = nullptr;
    uint64_t nslots = 0;
    int nsegs = 0;
    int64_t groups_capacity = 0;
    bool groups_compacted = false;     /* d_groups already holds final groups */
    unsigned col_null_flags[kMaxCols] = {0};
    uint64_t col_zzmin[kMaxCols] = {0};
    uint64_t col_zzmax[kMaxCols] = {0};
    hipStream_t stream = 0;

    ~DeviceRun()
    {
        g_pool.put(d_segs);
        g_pool.put(d_segex);
        g_pool.put(d_off);
        g_pool.put(d_cnt);
        g_pool.put(d_maxw);
        g_pool.put(d_err);
        g_pool.put(d_th);
        g_pool.put(d_slots);
        g_pool.put(d_groups);
        g_pool.put(d_counter);
        g_pool.put(d_gaccum);
        g_pool.put(d_fastcols);
        g_pool.put(d_colnull);
        g_pool.put(d_zzrange);
        g_pool.put(d_cursors);
        g_pool.put(d_recs);
        g_pool.put(d_ncursors);
        g_pool.put(d_nrecs);
    }
};

static uint64_t next_pow2(uint64_t x)
{
    uint64_t p = 1;
    while (p < x) p <<= 1;
    return p;
}

/* builds device segment descriptors; returns YT status.
 * input_row_limit > 0 truncates the scan to the first N rows — exactly the
 * reference's InputRowLimit interrupt semantics (rows are consumed in row
 * order; registry.cpp:259-265, TInterruptedIncompleteException). */
static int setup_chunk(const YtChunk* chunk, DeviceRun* R, unsigned* maxw_out,
                       int64_t input_row_limit, int* clamped,
                       char* errbuf, size_t errlen)
{
    int rc = YT_OK;
    int ncols = chunk->column_count;
    if (clamped) *clamped = 0;
    R->h_off.resize(ncols);
    R->h_cnt.resize(ncols);
    for (int c = 0; c < ncols; c++) {
        const YtColumn& col = chunk->columns[c];
        R->h_off[c] = (int32_t)R->h_segs.size();
        int32_t kept = 0;
        int64_t row = 0;
        for (int s = 0; s < col.segment_count; s++) {
            const YtSegment& seg = col.segments[s];
            int32_t rows_here = seg.row_count;
            if (input_row_limit > 0) {
                if (row >= input_row_limit) {
                    if (clamped) *clamped = 1;
                    row += seg.row_count;
                    continue;
                }
                if (row + rows_here > input_row_limit) {
                    rows_here = (int32_t)(input_row_limit - row);
                    if (clamped) *clamped = 1;
                }
            }
            DevSeg d;
            d.type = seg.type;
            d.is_signed = (col.value_type == YT_VT_STRING) ? 2
                        : (col.value_type == YT_VT_INT64);
            d.start_row = row;
            d.row_count = rows_here;
            d.col = c;
            d.min_value = seg.min_value;
            d.blob = (const uint64_t*)seg.data;
            d.blob_bytes = seg.data_size;
            R->h_segs.push_back(d);
            kept++;
            row += seg.row_count;
        }
        R->h_cnt[c] = kept;
        if (row != chunk->row_count) {
            set_err(errbuf, errlen, "segment row counts do not sum to chunk rows");
            return YT_ERR_INVALID_CHUNK;
        }
    }
    R->nsegs = (int)R->h_segs.size();
    if (R->nsegs == 0) return YT_OK;

    HIP_CHECK(pool_alloc(&R->d_segs, sizeof(DevSeg) * R->nsegs));
    HIP_CHECK(pool_alloc(&R->d_segex, sizeof(SegEx) * R->nsegs));
    HIP_CHECK(pool_alloc(&R->d_off, sizeof(int32_t) * ncols));
    HIP_CHECK(pool_alloc(&R->d_cnt, sizeof(int32_t) * ncols));
    HIP_CHECK(pool_alloc(&R->d_maxw, sizeof(unsigned)));
    HIP_CHECK(pool_alloc(&R->d_err, sizeof(unsigned)));
    HIP_CHECK(pool_alloc(&R->d_colnull, sizeof(unsigned) * kMaxCols));
    HIP_CHECK(hipMemsetAsync(R->d_colnull, 0, sizeof(unsigned) * kMaxCols, R->stream));
    HIP_CHECK(pool_alloc(&R->d_zzrange, sizeof(uint64_t) * 2 * kMaxCols));
    HIP_CHECK(hipMemsetAsync(R->d_zzrange, 0xFF, sizeof(uint64_t) * kMaxCols, R->stream));
    HIP_CHECK(hipMemsetAsync(R->d_zzrange + kMaxCols, 0, sizeof(uint64_t) * kMaxCols, R->stream));
    HIP_CHECK(hipMemcpyAsync(R->d_segs, R->h_segs.data(), sizeof(DevSeg) * R->nsegs,
                             hipMemcpyHostToDevice, R->stream));
    HIP_CHECK(hipMemcpyAsync(R->d_off, R->h_off.data(), sizeof(int32_t) * ncols,
                             hipMemcpyHostToDevice, R->stream));
    HIP_CHECK(hipMemcpyAsync(R->d_cnt, R->h_cnt.data(), sizeof(int32_t) * ncols,
                             hipMemcpyHostToDevice, R->stream));
    HIP_CHECK(hipMemsetAsync(R->d_maxw, 0, sizeof(unsigned), R->stream));
    HIP_CHECK(hipMemsetAsync(R->d_err, 0, sizeof(unsigned), R->stream));
    HIP_CHECK(ytql_launch_parse_segments(R->d_segs, R->nsegs, R->d_segex, R->d_maxw,
                                         R->d_colnull, R->d_zzrange,
                                         R->d_zzrange + kMaxCols, R->stream));
    HIP_CHECK(ytql_launch_scan_nullflags(R->d_segs, R->d_segex, R->nsegs,
                                         R->d_colnull, R->stream));
    HIP_CHECK(hipMemcpyAsync(maxw_out, R->d_maxw, sizeof(unsigned),
                             hipMemcpyDeviceToHost, R->stream));
    HIP_CHECK(hipMemcpyAsync(R->col_null_flags, R->d_colnull,
                             sizeof(unsigned) * kMaxCols,
                             hipMemcpyDeviceToHost, R->stream));
    HIP_CHECK(hipMemcpyAsync(R->col_zzmin, R->d_zzrange,
                             sizeof(uint64_t) * kMaxCols,
                             hipMemcpyDeviceToHost, R->stream));
    HIP_CHECK(hipMemcpyAsync(R->col_zzmax, R->d_zzrange + kMaxCols,
                             sizeof(uint64_t) * kMaxCols,
                             hipMemcpyDeviceToHost, R->stream));
    HIP_CHECK(hipStreamSynchronize(R->stream));
    return YT_OK;
fail:
    return rc;
}

static int setup_table(DeviceRun* R, int agg_count, int64_t max_groups,
                       int64_t group_limit, char* errbuf, size_t errlen)
{
    int rc = YT_OK;
    R->nslots = next_pow2((uint64_t)(max_groups > 0 ? max_groups : (1 << 20)) * 2);
    if (R->nslots < 2048) R->nslots = 2048;
    HIP_CHECK(pool_alloc(&R->d_th, sizeof(TableHdr)));
    TableHdr hh;
    memset(&hh, 0, sizeof(hh));
    hh.nslots = R->nslots;
    hh.mask = R->nslots - 1;
    hh.group_limit = group_limit;
    HIP_CHECK(hipMemcpyAsync(R->d_th, &hh, sizeof(hh), hipMemcpyHostToDevice, R->stream));
    return YT_OK;
fail:
    return rc;
}

/* the direct-table slot array is only needed on the non-partitioned paths */
static int ensure_slots(DeviceRun* R, int agg_count, char* errbuf, size_t errlen)
{
    int rc = YT_OK;
    if (R->d_slots) return YT_OK;
    int stride = 2 + 2 * agg_count;
    HIP_CHECK(pool_alloc(&R->d_slots, sizeof(uint64_t) * R->nslots * stride));
    HIP_CHECK(hipMemsetAsync(R->d_slots, 0, sizeof(uint64_t) * R->nslots * stride, R->stream));
    return YT_OK;
fail:
    return rc;
}

/* Two-phase partitioned group-by for the {1 direct int64 key, <=1 direct
 * sum, sum(1), optional range filter} family. On success sets *done and
 * leaves COMPACTED groups in R->d_groups (+ counter in R->d_counter);
 * sets *done=false (no error) when a capacity guard tripped so the caller
 * falls back to the direct-table kernel. */
static int run_partitioned(const YtPlan* plan, const YtChunk* chunk,
                           const YtExecOptions* options, DeviceRun* R,
                           const FastShape* fs, unsigned maxw,
                           YtStatistics* stats, bool* done,
                           int force_hash, bool* tried_direct,
                           char* errbuf, size_t errlen)
{
    int rc = YT_OK;
    *done = false;
    hipEvent_t ev0 = nullptr, ev1 = nullptr, ev2 = nullptr;

    PartParams pp;
    memset(&pp, 0, sizeof(pp));
    /* canonical slots: 0 = filter, 1 = key, 2 = value (see kernel comment) */
    int used[3];
    used[0] = fs->filter_col >= 0 ? fs->filter_col : fs->key_col;
    used[1] = fs->key_col;
    used[2] = fs->nsum == 1 ? fs->sum_col[0] : fs->key_col;
    int nused = 3;
    pp.filter_idx = fs->filter_col >= 0 ? 0 : -1;
    pp.key_idx = 1;
    pp.val_idx = fs->nsum == 1 ? 2 : -1;
    pp.nused = nused;
    pp.sum_slot = fs->nsum == 1 ? fs->sum_slot[0] : -1;
    pp.agg_count = plan->agg_count;
    pp.filter_lo = fs->filter_lo;
    pp.filter_hi = fs->filter_hi;
    pp.has_key_nulls = (R->col_null_flags[fs->key_col] != 0);
    pp.has_val_nulls = pp.val_idx >= 0 && (R->col_null_flags[fs->sum_col[0]] != 0);
    pp.has_filter_nulls = pp.filter_idx >= 0 && (R->col_null_flags[fs->filter_col] != 0);
    pp.stage_bm_mask = 0;
    for (int u = 0; u < nused; u++) {
        if (R->col_null_flags[used[u]] != 0) pp.stage_bm_mask |= 1 << u;
    }

    /* 8-byte packed records when the combined zigzag spans fit 64 bits */
    {
        auto bits_of = [](uint64_t span) -> int {
            int b = 0;
            while (span) { b++; span >>= 1; }
            return b;
        };
        auto exact_range = [&](int col) -> int {
            /* refine the meta-only span (rounded up to min+2^w-1 per
             * segment) by an exact device scan of the column */
            uint64_t* d_ex = nullptr;
            int rc2 = pool_alloc(&d_ex, 2 * sizeof(uint64_t));
            if (rc2 != YT_OK) return rc2;
            uint64_t init[2] = { ~0ULL, 0 };
            hipError_t e;
            if ((e = hipMemcpyAsync(d_ex, init, sizeof(init),
                                    hipMemcpyHostToDevice, R->stream)) ||
                (e = ytql_launch_scan_zzrange(R->d_segs, R->d_segex,
                                              R->h_off[col], R->h_cnt[col],
                                              R->col_null_flags[col] != 0,
                                              (unsigned long long*)d_ex, R->stream)) ||
                (e = hipMemcpyAsync(init, d_ex, sizeof(init),
                                    hipMemcpyDeviceToHost, R->stream)) ||
                (e = hipStreamSynchronize(R->stream))) {
                g_pool.put(d_ex);
                set_err(errbuf, errlen, hipGetErrorString(e));
                return YT_ERR_HIP;
            }
            g_pool.put(d_ex);
            if (init[0] <= init[1]) {   /* any non-null value seen */
                R->col_zzmin[col] = init[0];
                R->col_zzmax[col] = init[1];
            }
            return YT_OK;
        };
        uint64_t span_k = R->col_zzmax[fs->key_col] - R->col_zzmin[fs->key_col];
        int bk = bits_of(span_k);
        int bv = 0;
        uint64_t gmin_v = 0;
        if (pp.val_idx >= 0) {
            uint64_t span_v = R->col_zzmax[fs->sum_col[0]] - R->col_zzmin[fs->sum_col[0]];
            bv = bits_of(span_v);
            gmin_v = R->col_zzmin[fs->sum_col[0]];
        }
        /* the meta bound overstates each span by at most one bit; scan
         * exactly when that one bit could enable direct-span mode or the
         * aligned pad bit (cheap: one streaming read of the column) */
        if ((bk >= 1 && bk <= 23) && !(bk <= 22 && bk + bv <= 63)) {
            rc = exact_range(fs->key_col);
            if (rc != YT_OK) return rc;
            span_k = R->col_zzmax[fs->key_col] - R->col_zzmin[fs->key_col];
            bk = bits_of(span_k);
        }
        if (pp.val_idx >= 0 && bk + bv >= 64 && bk + bv <= 65) {
            rc = exact_range(fs->sum_col[0]);
            if (rc != YT_OK) return rc;
            uint64_t span_v = R->col_zzmax[fs->sum_col[0]] - R->col_zzmin[fs->sum_col[0]];
            bv = bits_of(span_v);
            gmin_v = R->col_zzmin[fs->sum_col[0]];
        }
        if (bk + bv <= 64) {
            pp.packed_mode = 1;
            pp.bits_k = bk ? bk : 1;
            if (pp.bits_k + bv > 64) pp.bits_k = bk;  /* bk>=1 here */
            pp.gmin_k = R->col_zzmin[fs->key_col];
            pp.gmin_v = gmin_v;
        }
        /* aligned 64B claims need a pad encoding: bit 63 for packed records
         * (spare iff bk+bv <= 63), kEmptyKey for 16B records (always).
         * Pads pay off only with the LDS reorder (coalesced full-line
         * writes); without it the 8B scatter's partial-line eviction isn't
         * fixed and pads just inflate the stream (measured r2c_ab). */
        int can_pad = (!pp.packed_mode || pp.bits_k + bv <= 63) ? 1 : 0;
        /* measured slower than the plain scatter at every tile size
         * (r2 A/Bs: 23.8-41 ms vs 18.7 ms) — L2 absorbs the 8B scatter
         * better than any LDS round-trip; keep as an experiment knob */
        pp.reorder = (pp.packed_mode && can_pad && !pp.has_val_nulls
                      && getenv("YTQL_REORDER")) ? 1 : 0;
        pp.aligned = pp.reorder;
        {
            const char* sm = getenv("YTQL_STORE");   /* perf experiments only */
            if (sm) pp.store_mode = atoi(sm);
        }
        /* measured ~1ms slower than XCD sub-streams (phase B loses its
         * paired-load ILP across 2048 short streams); keep as a knob */
        {
            const char* wgm = getenv("YTQL_WGSTREAMS");   /* 1 two-pass, 2 single-pass */
            pp.wg_streams = (!pp.reorder && !pp.has_val_nulls && wgm)
                ? atoi(wgm) : 0;
            if (pp.wg_streams == 2 && !pp.packed_mode) pp.wg_streams = 1;
        }
        /* direct-span mode: when the key zigzag span is small, partition by
         * key RANGE and index phase B's per-bucket array directly (the
         * headline config — 1M distinct keys — spans 21 bits). Needs the
         * pad encoding (phase B skips by sign bit), so requires aligned. */
        pp.gmin_k = R->col_zzmin[fs->key_col];
        if (bk <= 22 && can_pad && !force_hash && !getenv("YTQL_NO_DIRECT")) {
            pp.direct_mode = 1;
            pp.dshift = bk > 10 ? bk - 10 : 0;
            if (tried_direct) *tried_direct = true;
        }
    }

    int64_t rows = chunk->row_count;
    if (options->input_row_limit > 0 && options->input_row_limit < rows)
        rows = options->input_row_limit;
    int32_t seg0_rows = R->h_segs[R->h_off[fs->key_col]].row_count;

    /* LDS: 4*kNB u32 histogram structures + staged columns (value column's
     * packed words are NOT staged — read from global in the write pass) */
    unsigned w = maxw ? maxw : 1;
    /* bigger tiles make aligned claims cheaper (fewer pad tails per
     * bucket); bounded by LDS (key staging) at 2 workgroups/CU (<=80KB) */
    /* reorder: 31x256 rows -> scratch 62KB + 16KB histograms = 2 WGs/CU
     * (8192 would land 32 bytes over the 80KB half-LDS line).
     * Scatter mode: 4096 measured best (27KB LDS -> 5 WGs/CU; the kernel is
     * 73% WAIT_ANY, occupancy is the lever — r2 tile sweeps). */
    int tile_rows = pp.reorder ? 7936 : 4096;
    bool tile_forced = false;
    {
        const char* ev = getenv("YTQL_TILE");   /* perf experiments only */
        if (ev && atoi(ev) >= 256) { tile_rows = atoi(ev); tile_forced = true; }
        if (tile_rows > 16384) tile_rows = 16384;   /* per-thread row arrays bound */
    }
    /* LDS: per-tile bucket histograms + the staged key column */
    auto lds_for = [&](int tr) {
        /* reorder mode: record scratch instead of the staged key column;
         * wg single-pass: histograms only (keys read once, from global) */
        if (pp.reorder)
            return (size_t)tr * 8 + (size_t)4 * kNB * 4 + 64;
        if (pp.wg_streams == 2)
            return (size_t)4 * kNB * 4 + 256;
        return (size_t)4 * kNB * 4 + ((size_t)tr * w / 64 + 2) * 8 + 256;
    };
    size_t lds = lds_for(tile_rows);
    while (!tile_forced && lds > 80 * 1024 && tile_rows > 4096) { tile_rows >>= 1; lds = lds_for(tile_rows); }
    while (lds > 158 * 1024 && tile_rows > 1024) { tile_rows >>= 1; lds = lds_for(tile_rows); }
    while (tile_rows > 256 && tile_rows > seg0_rows) tile_rows >>= 1;
    lds = lds_for(tile_rows);
    pp.tile_rows = tile_rows;
    pp.tiles_per_seg = (seg0_rows + tile_rows - 1) / tile_rows;
    {
        int nseg = R->h_cnt[fs->key_col];
        int32_t last_rows = R->h_segs[R->h_off[fs->key_col] + nseg - 1].row_count;
        pp.ntiles = (nseg - 1) * pp.tiles_per_seg + (last_rows + tile_rows - 1) / tile_rows;
    }
    int grid = pp.ntiles < 2048 ? (pp.ntiles ? pp.ntiles : 1) : 2048;
    {
        const char* gv = getenv("YTQL_GRID");   /* perf experiments only */
        if (gv && atoi(gv) >= 8 && atoi(gv) < grid) grid = atoi(gv) & ~7;
        if (grid < 1) grid = 1;
    }
    int nsub = pp.wg_streams ? grid : 8;
    if (pp.wg_streams) {
        /* per-(bucket, WG) regions: rows this WG scans / kNB + skew slack */
        int64_t tiles_per_wg = (pp.ntiles + grid - 1) / grid;
        int64_t rows_per_wg = tiles_per_wg * pp.tile_rows;
        pp.bucket_stride = rows_per_wg / kNB + (rows_per_wg / kNB) / 2 + 64;
    } else {
        /* per (bucket, XCD) sub-streams: 8x more cursors, 1/8 the rows each.
         * Aligned mode adds up to 7 pad records per (tile,bucket,sub) claim —
         * budget the expected pad tail (~3.5 per claiming tile) explicitly. */
        pp.bucket_stride = rows / (kNB * 8) + (rows / (kNB * 8)) / 2 + 4096;
        if (pp.aligned) {
            int64_t tiles_per_sub = ((int64_t)pp.tiles_per_seg * R->h_cnt[fs->key_col] + 7) / 8;
            pp.bucket_stride += 4 * tiles_per_sub + 4096;
        }
    }
    /* every (bucket,sub) region must start 64B-aligned or aligned claims
     * land mid-line and every run straddles lines */
    pp.bucket_stride = (pp.bucket_stride + 7) & ~(int64_t)7;
    pp.nbucket_stride = pp.has_val_nulls ? pp.bucket_stride : 0;

    std::vector<FastCol> fc(nused);
    for (int u = 0; u < nused; u++) {
        fc[u].seg_off = R->h_off[used[u]];
        fc[u].seg_cnt = R->h_cnt[used[u]];
    }

    {
        HIP_CHECK(pool_alloc(&R->d_fastcols, sizeof(FastCol) * nused));
        HIP_CHECK(hipMemcpyAsync(R->d_fastcols, fc.data(), sizeof(FastCol) * nused,
                                 hipMemcpyHostToDevice, R->stream));
        HIP_CHECK(pool_alloc(&R->d_cursors, sizeof(uint64_t) * kNB * nsub));
        HIP_CHECK(hipMemsetAsync(R->d_cursors, 0, sizeof(uint64_t) * kNB * nsub, R->stream));
        HIP_CHECK(pool_alloc(&R->d_recs,
                             (size_t)kNB * nsub * pp.bucket_stride * (pp.packed_mode ? 8 : 16)));
        if (pp.has_val_nulls) {
            HIP_CHECK(pool_alloc(&R->d_ncursors, sizeof(uint64_t) * kNB * 8));
            HIP_CHECK(hipMemsetAsync(R->d_ncursors, 0, sizeof(uint64_t) * kNB * 8, R->stream));
            HIP_CHECK(pool_alloc(&R->d_nrecs, (size_t)kNB * 8 * pp.nbucket_stride * 8));
        }
        /* the partitioned path's in-table sentinel is INT64_MIN bits */
        uint64_t sk = kEmptyKey;
        HIP_CHECK(hipMemcpyAsync((char*)R->d_th + offsetof(TableHdr, side_key_bits),
                                 &sk, 8, hipMemcpyHostToDevice, R->stream));

        int64_t cap_groups = (int64_t)kNB * kHSlots;
        if (options->max_groups_hint > 0) {
            int64_t want = options->max_groups_hint * 2 + 4096;
            if (want < cap_groups) cap_groups = want;
        }
        if (cap_groups > rows + 16) cap_groups = rows + 16;
        R->groups_capacity = cap_groups;
        HIP_CHECK(pool_alloc(&R->d_groups, sizeof(OutGroup) * cap_groups));
        HIP_CHECK(pool_alloc(&R->d_counter, sizeof(unsigned long long)));
        HIP_CHECK(hipMemsetAsync(R->d_counter, 0, sizeof(unsigned long long), R->stream));

        HIP_CHECK(hipEventCreate(&ev0));
        HIP_CHECK(hipEventCreate(&ev1));
        HIP_CHECK(hipEventCreate(&ev2));
        HIP_CHECK(hipEventRecord(ev0, R->stream));
        HIP_CHECK(ytql_launch_scan_partition(&pp, R->d_segs, R->d_segex, R->d_fastcols,
                                             R->d_th, R->d_cursors, R->d_recs,
                                             R->d_ncursors, R->d_nrecs,
                                             lds, grid, R->stream));
        HIP_CHECK(hipEventRecord(ev1, R->stream));
        if (pp.direct_mode) {
            HIP_CHECK(ytql_launch_bucket_agg_direct(R->d_recs, R->d_cursors, pp.bucket_stride,
                                             R->d_nrecs, R->d_ncursors, pp.nbucket_stride,
                                             R->d_groups, R->d_counter, cap_groups,
                                             R->d_th, pp.sum_slot, pp.agg_count,
                                             pp.packed_mode, pp.bits_k,
                                             pp.gmin_k, pp.gmin_v, pp.dshift, nsub, R->stream));
        } else {
            HIP_CHECK(ytql_launch_bucket_agg(R->d_recs, R->d_cursors, pp.bucket_stride,
                                             R->d_nrecs, R->d_ncursors, pp.nbucket_stride,
                                             R->d_groups, R->d_counter, cap_groups,
                                             R->d_th, pp.sum_slot, pp.agg_count,
                                             pp.packed_mode, pp.bits_k,
                                             pp.gmin_k, pp.gmin_v, pp.aligned, nsub, R->stream));
        }
        HIP_CHECK(hipEventRecord(ev2, R->stream));
        HIP_CHECK(hipStreamSynchronize(R->stream));
        float msA = 0, msB = 0;
        HIP_CHECK(hipEventElapsedTime(&msA, ev0, ev1));
        HIP_CHECK(hipEventElapsedTime(&msB, ev1, ev2));
        if (stats) {
            stats->kernel_scan_ms += msA;
            stats->kernel_scan_launches += 1;
            stats->kernel_other_ms += msB;
        }
        /* overflow==1 → capacity guard; retry on the direct path */
        TableHdr th;
        HIP_CHECK(hipMemcpy(&th, R->d_th, sizeof(th), hipMemcpyDeviceToHost));
        if (th.overflow == 1) {
            /* reset and fall back */
            g_pool.put(R->d_groups); R->d_groups = nullptr;
            g_pool.put(R->d_counter); R->d_counter = nullptr;
            g_pool.put(R->d_recs); R->d_recs = nullptr;
            g_pool.put(R->d_cursors); R->d_cursors = nullptr;
            g_pool.put(R->d_nrecs); R->d_nrecs = nullptr;
            g_pool.put(R->d_ncursors); R->d_ncursors = nullptr;
            g_pool.put(R->d_fastcols); R->d_fastcols = nullptr;
            TableHdr hh;
            memset(&hh, 0, sizeof(hh));
            hh.nslots = R->nslots;
            hh.mask = R->nslots - 1;
            hh.group_limit = options->group_row_limit;
            HIP_CHECK(hipMemcpyAsync(R->d_th, &hh, sizeof(hh), hipMemcpyHostToDevice, R->stream));
            *done = false;
        } else {
            R->groups_compacted = true;
            *done = true;
        }
    }
    (void)hipEventDestroy(ev0);
    (void)hipEventDestroy(ev1);
    (void)hipEventDestroy(ev2);
    return YT_OK;
fail:
    if (ev0) (void)hipEventDestroy(ev0);
    if (ev1) (void)hipEventDestroy(ev1);
    if (ev2) (void)hipEventDestroy(ev2);
    return rc;
}

/* run the scan (fast when possible), leaving results in the table/gaccum.
 * Fills stats->kernel_scan_* from device events. */
static int run_scan(const YtPlan* plan, const YtChunk* chunk,
                    const YtExecOptions* options, DeviceRun* R,
                    const DevPlan* dp, const JoinDev* jd,
                    const JoinDev* jd2,
                    const FastShape* fs, unsigned maxw,
                    YtStatistics* stats, char* errbuf, size_t errlen)
{
    int rc = YT_OK;
    hipEvent_t ev0 = nullptr, ev1 = nullptr;

    if (fs->valid && fs->key_col >= 0 && fs->nsum <= 1) {
        bool done = false;
        bool tried_direct = false;
        rc = run_partitioned(plan, chunk, options, R, fs, maxw, stats, &done,
                             0, &tried_direct, errbuf, errlen);
        if (rc != YT_OK) return rc;
        /* direct-span mode overflows on range-clustered key skew; the
         * hash-partitioned layout spreads those — retry before giving up */
        if (!done && tried_direct) {
            rc = run_partitioned(plan, chunk, options, R, fs, maxw, stats, &done,
                                 1, nullptr, errbuf, errlen);
            if (rc != YT_OK) return rc;
        }
        if (done) return YT_OK;
        /* else: capacity guard tripped — fall through to the direct path */
    }

    rc = ensure_slots(R, plan->agg_count, errbuf, errlen);
    if (rc != YT_OK) return rc;

    HIP_CHECK(hipEventCreate(&ev0));
    HIP_CHECK(hipEventCreate(&ev1));

    HIP_CHECK(pool_alloc(&R->d_gaccum, sizeof(uint64_t) * (1 + 2 * kMaxAggs)));
    HIP_CHECK(hipMemsetAsync(R->d_gaccum, 0, sizeof(uint64_t) * (1 + 2 * kMaxAggs), R->stream));

    if (fs->valid) {
        /* fast fused kernel — canonical slots: 0 = filter, 1..4 = sum args,
         * 5 = key (see k_scan_fast) */
        FastParams fp;
        memset(&fp, 0, sizeof(fp));
        int used[6];
        used[0] = fs->filter_col >= 0 ? fs->filter_col : 0;
        for (int a = 0; a < kMaxAggs; a++) {
            used[1 + a] = (a < fs->nsum) ? fs->sum_col[a] : 0;
        }
        used[5] = fs->key_col >= 0 ? fs->key_col : 0;
        int nused = 6;
        fp.filter_idx = fs->filter_col >= 0 ? 0 : -1;
        fp.key_idx = fs->key_col >= 0 ? 5 : -1;
        fp.nsum = fs->nsum;
        for (int a = 0; a < fs->nsum; a++) {
            fp.sum_idx[a] = 1 + a;
            fp.sum_slot[a] = fs->sum_slot[a];
        }
        fp.nused = nused;
        fp.agg_count = plan->agg_count;
        fp.filter_lo = fs->filter_lo;
        fp.filter_hi = fs->filter_hi;
        fp.row_count = chunk->row_count;
        fp.nsegs_per_col = chunk->columns[0].segment_count;
        fp.stage_bm_mask = 0;
        if (fs->filter_col >= 0 && R->col_null_flags[fs->filter_col]) fp.stage_bm_mask |= 1;
        for (int a = 0; a < fs->nsum; a++) {
            if (R->col_null_flags[fs->sum_col[a]]) fp.stage_bm_mask |= 1 << (1 + a);
        }
        if (fs->key_col >= 0 && R->col_null_flags[fs->key_col]) fp.stage_bm_mask |= 1 << 5;

        int32_t seg0_rows = R->h_segs[R->h_off[used[1]]].row_count;

        /* no LDS staging: tile size only shapes the grid */
        size_t lds = 0;
        int tile_rows = 8192;
        if (tile_rows > seg0_rows) {
            while (tile_rows > 256 && tile_rows > seg0_rows) tile_rows >>= 1;
        }
        fp.tile_rows = tile_rows;
        fp.tiles_per_seg = (seg0_rows + tile_rows - 1) / tile_rows;
        {
            int nseg = R->h_cnt[used[1]];
            int32_t last_rows = R->h_segs[R->h_off[used[1]] + nseg - 1].row_count;
            fp.ntiles = (nseg - 1) * fp.tiles_per_seg + (last_rows + tile_rows - 1) / tile_rows;
        }

        std::vector<FastCol> fc(nused);
        for (int u = 0; u < nused; u++) {
            fc[u].seg_off = R->h_off[used[u]];
            fc[u].seg_cnt = R->h_cnt[used[u]];
        }
        HIP_CHECK(pool_alloc(&R->d_fastcols, sizeof(FastCol) * nused));
        HIP_CHECK(hipMemcpyAsync(R->d_fastcols, fc.data(), sizeof(FastCol) * nused,
                                 hipMemcpyHostToDevice, R->stream));

        int grid = fp.ntiles < 2048 ? (fp.ntiles ? fp.ntiles : 1) : 2048;
        HIP_CHECK(hipEventRecord(ev0, R->stream));
        HIP_CHECK(ytql_launch_scan_fast(&fp, R->d_segs, R->d_segex, R->d_fastcols,
                                        R->d_th, R->d_slots, R->d_gaccum,
                                        lds, grid, R->stream));
        HIP_CHECK(hipEventRecord(ev1, R->stream));
    } else {
        HIP_CHECK(hipEventRecord(ev0, R->stream));
        int64_t gen_rows = chunk->row_count;
        if (options->input_row_limit > 0 && options->input_row_limit < gen_rows)
            gen_rows = options->input_row_limit;
        HIP_CHECK(ytql_launch_scan_generic(dp, R->d_segs, R->d_segex, R->d_off, R->d_cnt,
                                           gen_rows, jd, jd2, R->d_th, R->d_slots,
                                           R->d_err, R->stream));
        HIP_CHECK(hipEventRecord(ev1, R->stream));
    }
    HIP_CHECK(hipStreamSynchronize(R->stream));
    {
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, ev0, ev1));
        if (stats) {
            stats->kernel_scan_ms += ms;
            stats->kernel_scan_launches += 1;
        }
    }
    if (!fs->valid) {
        unsigned kerr = 0;
        HIP_CHECK(hipMemcpy(&kerr, R->d_err, sizeof(unsigned), hipMemcpyDeviceToHost));
        if (kerr) {
            set_err(errbuf, errlen, kerr == YT_ERR_DIV_ZERO ? "Division by zero" : "expression error");
            rc = (int)kerr;
            goto fail;
        }
    }
    (void)hipEventDestroy(ev0);
    (void)hipEventDestroy(ev1);
    return YT_OK;
fail:
    if (ev0) (void)hipEventDestroy(ev0);
    if (ev1) (void)hipEventDestroy(ev1);
    return rc;
}

static inline int64_t h_zz_dec(uint64_t z)
{
    return (int64_t)(z >> 1) ^ -(int64_t)(z & 1);
}

/* resolve composite-key packing widths from the parsed per-column
 * zigzag-space ranges (k_parse_segments min/max atomics) */
static int pack_group_key(DevPlan* dp, const DeviceRun* R,
                          char* errbuf, size_t errlen)
{
    int shift = 0;
    for (int i = 0; i < dp->kp_count; i++) {
        int c = dp->kp_col[i];
        uint64_t lo = R->col_zzmin[c], hi = R->col_zzmax[c];
        uint64_t span = hi >= lo ? hi - lo : 0;
        /* codes are 0 (null) .. span+1: smallest bits with span+1 < 2^bits */
        int bits = 1;
        while (bits < 64 && ((span + 1) >> bits) != 0) bits++;
        dp->kp_base[i] = lo;
        dp->kp_bits[i] = bits;
        dp->kp_shift[i] = shift;
        shift += bits;
    }
    if (shift > 62) {
        set_err(errbuf, errlen,
                "multi-key GROUP BY: composite key wider than 62 bits this round");
        return YT_ERR_UNSUPPORTED;
    }
    return YT_OK;
}

/* finalize a group (OutGroup record or side accumulator) into HVal row
 * [keys..., aggs...] */
static void finalize_row(const YtPlan* plan, uint8_t key_type,
                         uint8_t sum_type[kMaxAggs],
                         uint64_t key_bits, int key_null, uint64_t cnt,
                         const uint64_t* agg_bits, const uint64_t* agg_nonnull,
                         HVal* row, int* nrow, int has_key)
{
    int n = 0;
    if (has_key) {
        row[n].type = key_null ? YT_VT_NULL : key_type;
        row[n].bits = key_null ? 0 : key_bits;
        n++;
    }
    for (int a = 0; a < plan->agg_count; a++) {
        if (plan->aggs[a]->func == YT_AGG_SUM1) {
            row[n].type = YT_VT_INT64;
            row[n].bits = cnt;
        } else if (agg_nonnull[a] == 0) {
            row[n].type = YT_VT_NULL;
            row[n].bits = 0;
        } else if (plan->aggs[a]->func == YT_AGG_AVG) {
            /* finalize: double(sum)/count (profiler avg Finalize; count —
             * the nonnull word — is nonzero here) */
            double sum;
            if (sum_type[a] == YT_VT_DOUBLE) {
                memcpy(&sum, &agg_bits[a], 8);
            } else if (sum_type[a] == YT_VT_UINT64) {
                sum = (double)agg_bits[a];
            } else {
                sum = (double)(int64_t)agg_bits[a];
            }
            double r = sum / (double)agg_nonnull[a];
            row[n].type = YT_VT_DOUBLE;
            memcpy(&row[n].bits, &r, 8);
        } else {
            row[n].type = sum_type[a];
            uint64_t bits = agg_bits[a];
            int f = plan->aggs[a]->func;
            if (f == YT_AGG_MIN || f == YT_AGG_MAX) {
                /* undo the order-preserving map (kernels.hip ord_map):
                 * MIN stored ~m(x); then m inverse per type */
                uint64_t m = (f == YT_AGG_MIN) ? ~bits : bits;
                const uint64_t SIGN = 0x8000000000000000ULL;
                if (sum_type[a] == YT_VT_DOUBLE) {
                    bits = (m & SIGN) ? (m & ~SIGN) : ~m;
                } else if (sum_type[a] == YT_VT_INT64) {
                    bits = m ^ SIGN;
                } else {
                    bits = m;
                }
            }
            row[n].bits = bits;
        }
        n++;
    }
    *nrow = n;
}

static int emit_rows(const YtPlan* plan, const YtChunk* chunk,
                     const DevPlan* dp,
                     const OutGroup* groups, int64_t ngroups,
                     const TableHdr& th, int has_any_row_global,
                     const uint64_t* gaccum, int used_fast_global,
                     int64_t output_row_limit, int* out_limited,
                     YtRowset* output, char* errbuf, size_t errlen)
{
    uint8_t col_types[kMaxCols];
    memset(col_types, YT_VT_INT64, sizeof(col_types));
    if (dp) {
        /* dp carries the EXTENDED column space (primary + joined foreign) */
        for (int c = 0; c < dp->ncols && c < kMaxCols; c++)
            col_types[c] = dp->col_types[c];
    } else {
        for (int c = 0; c < chunk->column_count && c < kMaxCols; c++)
            col_types[c] = (uint8_t)chunk->columns[c].value_type;
    }

    const int kp = dp ? dp->kp_count : 0;
    uint8_t key_type = YT_VT_INT64;
    int has_key = plan->key_count == 1;
    if (has_key) key_type = expr_static_type(plan->keys[0], col_types);
    uint8_t sum_type[kMaxAggs];
    for (int a = 0; a < plan->agg_count; a++) {
        sum_type[a] = (plan->aggs[a]->func != YT_AGG_SUM1)
            ? expr_static_type(plan->aggs[a]->arg, col_types) : YT_VT_INT64;
        if (sum_type[a] == YT_VT_NULL) sum_type[a] = YT_VT_INT64;
    }

    int base_cols = (kp ? kp : (has_key ? 1 : 0)) + plan->agg_count;
    int out_cols = plan->project_count ? plan->project_count : base_cols;

    auto emit_at = [&](YtValue* dst, uint64_t key_bits, int key_null,
                       uint64_t cnt, const uint64_t* ab, const uint64_t* an) -> int {
        HVal row[kMaxPackKeys + 1 + kMaxAggs];
        int nrow = 0;
        if (kp) {
            /* unpack the composite key (DevPlan kp_*; code 0 = null) */
            for (int i = 0; i < kp; i++) {
                uint64_t mask = dp->kp_bits[i] >= 64
                    ? ~0ULL : ((1ULL << dp->kp_bits[i]) - 1);
                uint64_t code = (key_bits >> dp->kp_shift[i]) & mask;
                if (code == 0) {
                    row[nrow].type = YT_VT_NULL;
                    row[nrow].bits = 0;
                } else {
                    uint64_t z = code - 1 + dp->kp_base[i];
                    row[nrow].type = col_types[dp->kp_col[i]];
                    row[nrow].bits = dp->kp_signed[i]
                        ? (uint64_t)h_zz_dec(z) : z;
                }
                nrow++;
            }
            int an_ = 0;
            finalize_row(plan, key_type, sum_type, 0, 0, cnt, ab, an,
                         row + nrow, &an_, /*has_key=*/0);
            nrow += an_;
        } else
        finalize_row(plan, key_type, sum_type, key_bits, key_null, cnt, ab, an,
                     row, &nrow, has_key);
        if (plan->project_count) {
            for (int p = 0; p < plan->project_count; p++) {
                HVal v;
                int rc = heval(plan->projects[p], row, nrow, &v);
                if (rc) return rc;
                dst[p].id = (uint16_t)p;
                dst[p].type = v.type;
                dst[p].flags = 0;
                dst[p].length = 0;
                dst[p].data.bits = v.bits;
            }
        } else {
            for (int i = 0; i < nrow; i++) {
                dst[i].id = (uint16_t)i;
                dst[i].type = row[i].type;
                dst[i].flags = 0;
                dst[i].length = 0;
                dst[i].data.bits = row[i].bits;
            }
        }
        return YT_OK;
    };
    auto emit = [&](uint64_t key_bits, int key_null, uint64_t cnt,
                    const uint64_t* ab, const uint64_t* an) -> int {
        if (output_row_limit > 0 && output->row_count >= output_row_limit) {
            /* OutputRowLimit: soft stop — registry.cpp:297-305 WriteRow */
            if (out_limited) *out_limited = 1;
            return -1;
        }
        if (output->row_count >= output->capacity_rows) return YT_ERR_CAPACITY;
        int rc_ = emit_at(output->values + output->row_count * out_cols,
                          key_bits, key_null, cnt, ab, an);
        if (rc_ == YT_OK) output->row_count++;
        return rc_;
    };

    output->row_count = 0;
    output->column_count = out_cols;
    int rc = YT_OK;

    if (!has_key && !kp) {
        /* global aggregate: one row iff any row passed the filter */
        uint64_t cnt;
        uint64_t ab[kMaxAggs] = {0}, an[kMaxAggs] = {0};
        if (used_fast_global) {
            cnt = gaccum[0];
            for (int a = 0; a < plan->agg_count; a++) {
                ab[a] = gaccum[1 + 2 * a];
                an[a] = gaccum[2 + 2 * a];
            }
            if (cnt == 0) return YT_OK;
        } else {
            /* generic path routed rows to synthetic key bits=1 */
            if (ngroups == 0) return YT_OK;
            cnt = groups[0].cnt;
            for (int a = 0; a < plan->agg_count; a++) {
                ab[a] = groups[0].agg_bits[a];
                an[a] = groups[0].agg_nonnull[a];
            }
        }
        (void)has_any_row_global;
        rc = emit(0, 0, cnt, ab, an);
        return rc == -1 ? YT_OK : rc;
    }

    {
        const int64_t total_sides = (int64_t)(th.side_used[0] != 0)
                                  + (int64_t)(th.side_used[1] != 0);
        const bool par_ok = ngroups >= 65536
            && (output_row_limit <= 0 ||
                output_row_limit >= ngroups + total_sides)
            && ngroups + total_sides <= output->capacity_rows;
        if (par_ok) {
            /* row i is group i — emit in parallel (group order preserved;
             * the host loop was 5.5 ms of the 31 ms step at 1M groups) */
            std::atomic<int> arc{YT_OK};
            int nt = (int)std::min<int64_t>(std::thread::hardware_concurrency(),
                                            (ngroups + 131071) / 131072);
            if (nt > 32) nt = 32;
            if (nt < 1) nt = 1;
            std::vector<std::thread> ths;
            for (int t = 0; t < nt; t++) {
                ths.emplace_back([&, t]() {
                    int64_t lo = ngroups * t / nt, hi = ngroups * (t + 1) / nt;
                    for (int64_t i = lo; i < hi; i++) {
                        const OutGroup& g = groups[i];
                        int r2 = emit_at(output->values + i * out_cols,
                                         g.key_bits, (int)(g.key_meta & 1),
                                         g.cnt, g.agg_bits, g.agg_nonnull);
                        if (r2 != YT_OK) { arc.store(r2); return; }
                    }
                });
            }
            for (auto& t : ths) t.join();
            if (arc.load() != YT_OK) return arc.load();
            output->row_count = ngroups;
        } else {
            for (int64_t i = 0; i < ngroups; i++) {
                const OutGroup& g = groups[i];
                rc = emit(g.key_bits, (int)(g.key_meta & 1), g.cnt,
                          g.agg_bits, g.agg_nonnull);
                if (rc == -1) return YT_OK;
                if (rc) return rc;
            }
        }
    }
    /* side groups: the in-table sentinel key, then the null key */
    for (int side = 0; side < 2; side++) {
        if (!th.side_used[side]) continue;
        uint64_t ab[kMaxAggs], an[kMaxAggs];
        for (int a = 0; a < plan->agg_count; a++) {
            ab[a] = th.side_agg[side][2 * a];
            an[a] = th.side_agg[side][2 * a + 1];
        }
        rc = emit(th.side_key_bits[side], side == 1, th.side_cnt[side], ab, an);
        if (rc == -1) return YT_OK;
        if (rc) return rc;
    }
    (void)errbuf; (void)errlen;
    return rc;
}


/* ---- ORDER BY ... LIMIT ----
 * Comparer mirrors the reference codegen universal comparer
 * (cg_fragment_compiler.cpp:400-530): null < any, int64 signed,
 * uint64/boolean unsigned, double by value (NaN comparison = error), string
 * memcmp + length tiebreak; a descending key inverts the outcome. The
 * collector contract is TTopCollector's (top_collector-inl.h AddRow +
 * OrderOpHelper registry.cpp:1948-1997): keep the (offset+limit) least
 * rows, emit sorted ascending from offset. */
static int ord_cmp_vals(const YtPlan* plan, const YtValue* a, const YtValue* b,
                        int* nan_err)
{
    for (int i = 0; i < plan->order_count; i++) {
        int c = plan->order_cols[i];
        const YtValue* x = &a[c];
        const YtValue* y = &b[c];
        int xn = x->type == YT_VT_NULL, yn = y->type == YT_VT_NULL;
        int r = 0;
        if (xn || yn) {
            r = (xn == yn) ? 0 : (xn ? -1 : 1);
        } else if (x->type == YT_VT_DOUBLE) {
            double xv = x->data.dbl, yv = y->data.dbl;
            if (xv != xv || yv != yv) { *nan_err = 1; return 0; }
            r = xv < yv ? -1 : (xv > yv ? 1 : 0);
        } else if (x->type == YT_VT_STRING) {
            uint32_t lx = x->length, ly = y->length, m = lx < ly ? lx : ly;
            int mc = m ? memcmp(x->data.str, y->data.str, m) : 0;
            r = mc ? (mc < 0 ? -1 : 1) : (lx < ly ? -1 : (lx > ly ? 1 : 0));
        } else if (x->type == YT_VT_INT64) {
            r = x->data.i64 < y->data.i64 ? -1 : (x->data.i64 > y->data.i64 ? 1 : 0);
        } else {
            r = x->data.u64 < y->data.u64 ? -1 : (x->data.u64 > y->data.u64 ? 1 : 0);
        }
        if (plan->order_desc && plan->order_desc[i]) r = -r;
        if (r) return r;
    }
    return 0;
}

static int ord_validate_plan(const YtPlan* plan, int ncols, char* errbuf,
                             size_t errlen)
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

/* sort the already-materialized rowset and keep [offset, offset+limit) —
 * used for grouped output, where the group count is already bounded */
static int apply_order_host(const YtPlan* plan, YtRowset* out,
                            char* errbuf, size_t errlen)
{
    int ncols = out->column_count;
    int rc = ord_validate_plan(plan, ncols, errbuf, errlen);
    if (rc != YT_OK) return rc;
    int64_t n = out->row_count;
    std::vector<const YtValue*> idx((size_t)n);
    for (int64_t i = 0; i < n; i++) idx[i] = out->values + i * ncols;
    int nan_err = 0;
    std::sort(idx.begin(), idx.end(),
              [&](const YtValue* a, const YtValue* b) {
                  return ord_cmp_vals(plan, a, b, &nan_err) < 0;
              });
    if (nan_err) {
        set_err(errbuf, errlen, "NaN in ORDER BY comparison");
        return YT_ERR_LIMIT;
    }
    int64_t b = plan->order_offset < n ? plan->order_offset : n;
    int64_t e = b + plan->order_limit;
    if (e > n) e = n;
    std::vector<YtValue> tmp((size_t)(e - b) * ncols);
    for (int64_t i = b; i < e; i++)
        memcpy(tmp.data() + (i - b) * ncols, idx[i], sizeof(YtValue) * ncols);
    memcpy(out->values, tmp.data(), sizeof(YtValue) * (e - b) * ncols);
    out->row_count = e - b;
    return YT_OK;
}

/* WITH TOTALS + ORDER BY finishing pass over a fully-materialized group
 * rowset: totals (null keys + aggregates over ALL grouped rows — the
 * reference's EStreamTag::Totals, registry.cpp FlushTotals) are computed
 * BEFORE the order/limit slice and appended after it, flagged in
 * YtRowset.totals_row. */
static int column_uniform_shift(const YtColumn& col)
{
    if (col.segment_count == 0) return 0;
    int32_t r0 = col.segments[0].row_count;
    if (r0 <= 0 || (r0 & (r0 - 1)) != 0) return 0;
    for (int i = 0; i + 1 < col.segment_count; i++)
        if (col.segments[i].row_count != r0) return 0;
    if (col.segments[col.segment_count - 1].row_count > r0) return 0;
    int shift = 0;
    while ((1 << (shift + 1)) <= r0) shift++;
    return (1 << shift) == r0 ? shift : 0;
}

/* equi-join runtime: parsed foreign chunk + unique-key hash table in HBM */
struct JoinRun {
    DeviceRun Rf;
    JoinDev jd;
    uint64_t* d_hkey = nullptr;
    long long* d_hrow = nullptr;
    unsigned long long* d_misc = nullptr;

    long long* d_chead = nullptr;
    long long* d_fnext = nullptr;

    JoinRun() { memset(&jd, 0, sizeof(jd)); }
    ~JoinRun()
    {
        g_pool.put(d_hkey);
        g_pool.put(d_hrow);
        g_pool.put(d_misc);
        g_pool.put(d_chead);
        g_pool.put(d_fnext);
    }
};

/* one join item: the primary side is the row AS EXTENDED SO FAR
 * (row_types/row_ncols — primary columns plus earlier items' values, so a
 * later item can key on an earlier item's output). */
static int setup_join_item(const YtJoin* J, const uint8_t* row_types,
                           int row_ncols, JoinRun* JR,
                           hipStream_t stream, char* errbuf, size_t errlen)
{
    int rc = YT_OK;
    const YtChunk* fc = J->foreign;
    if (J->primary_key_col < 0 || J->primary_key_col >= row_ncols ||
        J->foreign_key_col < 0 || J->foreign_key_col >= fc->column_count) {
        set_err(errbuf, errlen, "join: key column out of range");
        return YT_ERR_INVALID_PLAN;
    }
    auto intish = [](int vt) {
        return vt == YT_VT_INT64 || vt == YT_VT_UINT64 || vt == YT_VT_BOOLEAN;
    };
    if (!intish(row_types[J->primary_key_col]) ||
        !intish(fc->columns[J->foreign_key_col].value_type)) {
        set_err(errbuf, errlen,
                "join: int64/uint64/boolean key columns this round");
        return YT_ERR_UNSUPPORTED;
    }
    for (int j = 0; j < J->foreign_value_count; j++) {
        int cjf = J->foreign_value_cols[j];
        if (cjf < 0 || cjf >= fc->column_count) {
            set_err(errbuf, errlen, "join: foreign value column out of range");
            return YT_ERR_INVALID_PLAN;
        }
        if (fc->columns[cjf].value_type == YT_VT_STRING) {
            set_err(errbuf, errlen, "join: string foreign values not this round");
            return YT_ERR_UNSUPPORTED;
        }
    }

    JR->Rf.stream = stream;
    unsigned mw = 0;
    int cl = 0;
    rc = setup_chunk(fc, &JR->Rf, &mw, 0, &cl, errbuf, errlen);
    if (rc) return rc;

    int64_t fn = fc->row_count;
    uint64_t cap = next_pow2((uint64_t)(fn ? fn : 1) * 2);
    if (cap < 2048) cap = 2048;
    JoinDev& jd = JR->jd;
    unsigned long long nr1 = 0;
    unsigned kerr = 0;
    HIP_CHECK(pool_alloc(&JR->d_hkey, sizeof(uint64_t) * cap));
    HIP_CHECK(pool_alloc(&JR->d_hrow, sizeof(long long) * cap));
    HIP_CHECK(pool_alloc(&JR->d_misc, sizeof(unsigned long long) * 2));
    HIP_CHECK(pool_alloc(&JR->d_chead, sizeof(long long) * cap));
    HIP_CHECK(pool_alloc(&JR->d_fnext, sizeof(long long) * (fn ? fn : 1)));
    HIP_CHECK(hipMemsetAsync(JR->d_hrow, 0xFF, sizeof(long long) * cap, stream));
    HIP_CHECK(hipMemsetAsync(JR->d_chead, 0xFF, sizeof(long long) * cap, stream));
    HIP_CHECK(hipMemsetAsync(JR->d_fnext, 0xFF, sizeof(long long) * (fn ? fn : 1), stream));
    HIP_CHECK(hipMemsetAsync(JR->d_misc, 0, sizeof(unsigned long long) * 2, stream));

    jd.active = 1;
    jd.is_left = J->is_left ? 1 : 0;
    jd.pkey_col = J->primary_key_col;
    jd.primary_ncols = row_ncols;
    jd.fkey_col = J->foreign_key_col;
    jd.fkey_shift = column_uniform_shift(fc->columns[J->foreign_key_col]);
    for (int j = 0; j < J->foreign_value_count; j++) {
        jd.fval_col[j] = J->foreign_value_cols[j];
        jd.f_shift[j] = column_uniform_shift(fc->columns[J->foreign_value_cols[j]]);
    }
    jd.fsegs = JR->Rf.d_segs;
    jd.fsegex = JR->Rf.d_segex;
    jd.f_off = JR->Rf.d_off;
    jd.f_cnt = JR->Rf.d_cnt;
    jd.hkey = JR->d_hkey;
    jd.hrow = (const int64_t*)JR->d_hrow;
    jd.hmask = cap - 1;
    jd.null_row = -1;
    jd.frows = fn;
    jd.chead = (const int64_t*)JR->d_chead;
    jd.fnext = (const int64_t*)JR->d_fnext;
    jd.has_dups = 0;

    if (fn > 0 && JR->Rf.nsegs > 0) {
        /* d_err[0] = decode error, d_err[1] = has_dups flag (two words) */
        HIP_CHECK(hipMemsetAsync(JR->Rf.d_err, 0, sizeof(unsigned), stream));
        HIP_CHECK(hipMemsetAsync(JR->d_misc + 1, 0, sizeof(unsigned long long), stream));
        HIP_CHECK(ytql_launch_join_build(&jd, fn, JR->d_hkey, JR->d_hrow,
                                         JR->d_misc, (unsigned*)(JR->d_misc + 1), stream));
        HIP_CHECK(hipMemcpy(&nr1, JR->d_misc, sizeof(unsigned long long),
                            hipMemcpyDeviceToHost));
        jd.null_row = nr1 ? (int64_t)(nr1 - 1) : -1;
        HIP_CHECK(ytql_launch_join_chain(&jd, fn, (unsigned*)(JR->d_misc + 1), stream));
        unsigned long long dupw = 0;
        HIP_CHECK(hipMemcpy(&dupw, JR->d_misc + 1, sizeof(unsigned long long),
                            hipMemcpyDeviceToHost));
        HIP_CHECK(hipMemcpy(&kerr, JR->Rf.d_err, sizeof(unsigned),
                            hipMemcpyDeviceToHost));
        jd.has_dups = ((unsigned*)&dupw)[1] != 0;
        if (kerr) {
            set_err(errbuf, errlen, "join: foreign decode error");
            return (int)kerr;
        }
        HIP_CHECK(hipMemsetAsync(JR->Rf.d_err, 0, sizeof(unsigned), stream));
    }
    return YT_OK;
fail:
    return rc;
}


/* drive the join chain (2 items max this round; duplicate foreign keys on
 * the FIRST item only — the cross-product machinery binds item 0) */
static int setup_join_chain(const YtPlan* plan, const YtChunk* chunk,
                            JoinRun* JR, JoinDev const** jd0,
                            JoinDev const** jd1,
                            hipStream_t stream, char* errbuf, size_t errlen)
{
    static const JoinDev kNone = {};
    *jd0 = &kNone;
    *jd1 = &kNone;
    uint8_t rt[kMaxCols];
    int rn = chunk->column_count;
    if (rn > kMaxCols) { set_err(errbuf, errlen, "too many columns"); return YT_ERR_UNSUPPORTED; }
    for (int c = 0; c < rn; c++) rt[c] = (uint8_t)chunk->columns[c].value_type;
    int nj = 0;
    for (const YtJoin* J = plan->join; J; J = J->next) {
        if (nj >= 2) {
            set_err(errbuf, errlen, "join: more than two join items not this round");
            return YT_ERR_UNSUPPORTED;
        }
        int rc = setup_join_item(J, rt, rn, &JR[nj], stream, errbuf, errlen);
        if (rc != YT_OK) return rc;
        if (nj == 1 && JR[1].jd.has_dups) {
            set_err(errbuf, errlen,
                    "join: duplicate foreign keys on a non-first join item "
                    "not this round");
            return YT_ERR_UNSUPPORTED;
        }
        for (int j = 0; j < J->foreign_value_count; j++) {
            if (rn >= kMaxCols) { set_err(errbuf, errlen, "too many columns"); return YT_ERR_UNSUPPORTED; }
            rt[rn++] = (uint8_t)J->foreign->columns[J->foreign_value_cols[j]].value_type;
        }
        nj++;
    }
    if (nj >= 1) *jd0 = &JR[0].jd;
    if (nj >= 2) *jd1 = &JR[1].jd;
    return YT_OK;
}

static void fold_totals_rows(const YtPlan* plan, const YtRowset* out,
                             std::vector<YtValue>& tot)
{
    int kc = plan->key_count, ac = plan->agg_count;
    int ncols = out->column_count;
    tot.assign(ac, YtValue());
    for (int a = 0; a < ac; a++) {
        tot[a].id = (uint16_t)(kc + a);
        tot[a].type = YT_VT_NULL;
    }
    for (int64_t r = 0; r < out->row_count; r++) {
        const YtValue* row = out->values + r * ncols;
        for (int a = 0; a < ac; a++) {
            const YtValue& v = row[kc + a];
            if (v.type == YT_VT_NULL) continue;
            YtValue& t = tot[a];
            int f = plan->aggs[a]->func;
            if (t.type == YT_VT_NULL) {
                t.type = v.type;
                t.data.bits = v.data.bits;
                continue;
            }
            if (f == YT_AGG_SUM || f == YT_AGG_SUM1) {
                if (v.type == YT_VT_DOUBLE) t.data.dbl += v.data.dbl;
                else t.data.bits += v.data.bits;   /* mod 2^64, udf/sum.c */
            } else if (f == YT_AGG_FIRST) {
                /* keep the first non-null (already in t) */
            } else {
                bool take;
                if (v.type == YT_VT_DOUBLE)
                    take = (f == YT_AGG_MAX) ? (v.data.dbl > t.data.dbl)
                                             : (v.data.dbl < t.data.dbl);
                else if (v.type == YT_VT_INT64)
                    take = (f == YT_AGG_MAX) ? (v.data.i64 > t.data.i64)
                                             : (v.data.i64 < t.data.i64);
                else
                    take = (f == YT_AGG_MAX) ? (v.data.u64 > t.data.u64)
                                             : (v.data.u64 < t.data.u64);
                if (take) t.data.bits = v.data.bits;
            }
        }
    }
}

static void hexpr_cols(const YtExpr* e, uint64_t* mask)
{
    if (!e) return;
    if (e->op == YT_EX_COLUMN && e->col >= 0 && e->col < 64)
        *mask |= 1ULL << e->col;
    hexpr_cols(e->a, mask);
    hexpr_cols(e->b, mask);
}

/* totals(BeforeHaving) → HAVING → totals(AfterHaving) → ORDER/limit →
 * append totals row (folding_profiler.cpp:1810-1815 Process() order;
 * parser.ypp:469-481 totals modes) */
static int finish_output(const YtPlan* plan, YtRowset* out,
                         char* errbuf, size_t errlen)
{
    std::vector<YtValue> tot;
    const bool after = plan->totals_mode == 2;
    /* all-null group keys are forbidden on the INTERMEDIATE stream (WITH
     * TOTALS re-folds rows) and in the group-combined-with-order op —
     * registry.cpp ValidateGroupKeyIsNotNull:1460-1476, call sites
     * :1795,:1820; pinned by GroupByWithTotalsNulls ql_query_ut.cpp:3989 */
    if (plan->with_totals || plan->order_count > 0) {
        const int kc0 = plan->key_count;
        const int nc0 = out->column_count;
        for (int64_t r = 0; r < out->row_count; r++) {
            const YtValue* row = out->values + r * nc0;
            bool allnull = kc0 > 0;
            for (int k = 0; k < kc0; k++)
                if (row[k].type != YT_VT_NULL) { allnull = false; break; }
            if (allnull) {
                set_err(errbuf, errlen, "Null values are forbidden in group key");
                return YT_ERR_INVALID_PLAN;
            }
        }
    }
    const int64_t had_group_rows = out->row_count;
    if (plan->with_totals && !after) fold_totals_rows(plan, out, tot);
    if (plan->having) {
        int ncols = out->column_count;
        uint64_t mask = 0;
        hexpr_cols(plan->having, &mask);
        int64_t w = 0;
        HVal hrow[32];
        for (int64_t r = 0; r < out->row_count; r++) {
            YtValue* row = out->values + r * ncols;
            for (int i = 0; i < ncols && i < 32; i++) {
                if (((mask >> i) & 1) && row[i].type == YT_VT_STRING) {
                    set_err(errbuf, errlen,
                            "HAVING over string values: not this round");
                    return YT_ERR_UNSUPPORTED;
                }
                hrow[i].type = row[i].type;
                hrow[i].bits = row[i].data.bits;
            }
            HVal h;
            int rc = heval(plan->having, hrow, ncols, &h);
            if (rc) { set_err(errbuf, errlen, "HAVING expression error"); return rc; }
            if (h.type == YT_VT_NULL || h.bits == 0) continue;
            if (w != r)
                memmove(out->values + w * ncols, row, sizeof(YtValue) * ncols);
            w++;
        }
        out->row_count = w;
    }
    if (plan->with_totals && after) fold_totals_rows(plan, out, tot);
    if (plan->order_count > 0) {
        int rc = apply_order_host(plan, out, errbuf, errlen);
        if (rc) return rc;
    }
    if (plan->with_totals && had_group_rows > 0) {
        int kc = plan->key_count, ac = plan->agg_count;
        int ncols = out->column_count;
        if (out->row_count >= out->capacity_rows) {
            set_err(errbuf, errlen, "rowset too small for the totals row");
            return YT_ERR_CAPACITY;
        }
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
            if (plan->aggs[a]->func == YT_AGG_SUM1 &&
                dst[kc + a].type == YT_VT_NULL) {
                dst[kc + a].type = YT_VT_INT64;   /* sum(1) over zero rows */
                dst[kc + a].data.bits = 0;
            }
        }
        out->row_count++;
        out->totals_row = 1;
    }
    return YT_OK;
}

/* scan + ORDER BY ... LIMIT without materializing the scan: histogram
 * k-selection on an order-isomorphic u64 mapping of the first order key
 * (digits of 11 bits, host-driven refinement), then exact candidate gather
 * and a full-comparer host sort of the <=(K + tie-cap) survivors.
 * Selection runs the same expression program as the projection, so any
 * filter/projection the generic path supports works here. */
static int run_scan_topk(const YtPlan* plan, const YtChunk* chunk,
                         const YtExecOptions* options, const DevPlan* dp,
                         const JoinDev* jd, const JoinDev* jd2,
                         YtRowset* output, YtStatistics* stats, double tw0,
                         char* errbuf, size_t errlen)
{
    int rc = YT_OK;
    const int np = plan->project_count;
    rc = ord_validate_plan(plan, np, errbuf, errlen);
    if (rc != YT_OK) return rc;
    const int64_t K = plan->order_offset + plan->order_limit;
    if (K > ((int64_t)1 << 24)) {
        set_err(errbuf, errlen, "ORDER BY LIMIT capped at 16M rows this round");
        return YT_ERR_UNSUPPORTED;
    }

    DeviceRun R2;
    R2.stream = (hipStream_t)(uintptr_t)options->stream;
    unsigned mw = 0;
    int in_clamped = 0;
    rc = setup_chunk(chunk, &R2, &mw, options->input_row_limit, &in_clamped,
                     errbuf, errlen);
    if (rc) return rc;
    if (stats) stats->decode_time_ms = now_ms() - tw0;   /* setup phase */
    int64_t n = chunk->row_count;
    if (options->input_row_limit > 0 && options->input_row_limit < n)
        n = options->input_row_limit;
    output->row_count = 0;
    output->column_count = np;
    if (n == 0 || R2.nsegs == 0) return YT_OK;
    if (in_clamped && stats) stats->incomplete_input = 1;

    const int ord0 = plan->order_cols[0];
    const int desc0 = plan->order_desc ? plan->order_desc[0] : 0;
    const int64_t CAPB = (int64_t)1 << 17;

    /* fast selection passes: plain DirectDense int column key, no filter,
     * no join, uniform segmentation (k_topk_*_fast — 4 rows in flight per
     * thread, no interpreter) */
    int fast_col = -1, fast_signed = 0, fast_nulls = 0, fast_shift = 0;
    {
        int kcol;
        if (!plan->filter && !plan->join &&
            expr_is_col(plan->projects[ord0], &kcol) &&
            kcol < chunk->column_count &&
            (chunk->columns[kcol].value_type == YT_VT_INT64 ||
             chunk->columns[kcol].value_type == YT_VT_UINT64) &&
            dp->col_uniform_shift[kcol] != 0) {
            bool all_dd = true;
            for (int i = 0; i < chunk->columns[kcol].segment_count; i++) {
                if (chunk->columns[kcol].segments[i].type != YT_SEG_DIRECT_DENSE) {
                    all_dd = false;
                    break;
                }
            }
            if (all_dd) {
                fast_col = kcol;
                fast_signed = chunk->columns[kcol].value_type == YT_VT_INT64;
                fast_nulls = R2.col_null_flags[kcol] != 0;
                fast_shift = dp->col_uniform_shift[kcol];
            }
        }
    }

    {
        unsigned long long* d_bins = nullptr;   /* 2048 bins + misc(3) + ctrs(3) */
        HIP_CHECK(pool_alloc(&d_bins, sizeof(unsigned long long) * (2048 + 8)));
        unsigned long long* d_misc = d_bins + 2048;   /* null_cnt, mmin, mmax */
        unsigned long long* d_ctrs = d_bins + 2051;   /* strict, tie, null */
        HIP_CHECK(hipMemsetAsync(R2.d_err, 0, sizeof(unsigned), R2.stream));

        hipEvent_t e0, e1;
        HIP_CHECK(hipEventCreate(&e0));
        HIP_CHECK(hipEventCreate(&e1));
        HIP_CHECK(hipEventRecord(e0, R2.stream));

        TopkPass tp;
        tp.order_proj = ord0;
        tp.desc = desc0;
        tp.level0 = 1;
        tp.shift = 53;
        tp.lo = 0;
        tp.hi = ~0ULL;

        uint64_t lo = 0, hi = ~0ULL;
        int shift = 53;
        int64_t S = 0, T = 0, nonnull = 0, k_nonnull = 0;
        unsigned long long null_cnt = 0;
        bool all_nonnull = false, only_nulls = false;
        std::vector<unsigned long long> h_bins(2051);
        int launches = 0;

        for (;;) {
            HIP_CHECK(hipMemsetAsync(d_bins, 0,
                                     sizeof(unsigned long long) * 2049, R2.stream));
            if (tp.level0) {
                /* mmin = ~0, mmax = 0 */
                HIP_CHECK(hipMemsetAsync(d_misc + 1, 0xFF,
                                         sizeof(unsigned long long), R2.stream));
                HIP_CHECK(hipMemsetAsync(d_misc + 2, 0,
                                         sizeof(unsigned long long), R2.stream));
            }
            if (fast_col >= 0) {
                HIP_CHECK(ytql_launch_topk_hist_fast(
                    R2.d_segs, R2.d_segex, R2.h_off[fast_col], fast_shift, n,
                    fast_nulls, fast_signed, &tp, d_bins, d_misc, R2.stream));
            } else {
                HIP_CHECK(ytql_launch_topk_hist(dp, R2.d_segs, R2.d_segex,
                                                R2.d_off, R2.d_cnt, n, jd,
                                                jd2, &tp,
                                                d_bins, d_misc,
                                                R2.d_err, R2.stream));
            }
            launches++;
            HIP_CHECK(hipMemcpy(h_bins.data(), d_bins,
                                sizeof(unsigned long long) * 2051,
                                hipMemcpyDeviceToHost));
            unsigned kerr = 0;
            HIP_CHECK(hipMemcpy(&kerr, R2.d_err, sizeof(unsigned),
                                hipMemcpyDeviceToHost));
            if (kerr) {
                g_pool.put(d_bins);
                if (kerr == 100) {
                    set_err(errbuf, errlen, "NaN in ORDER BY comparison");
                    return YT_ERR_LIMIT;
                }
                set_err(errbuf, errlen, "unsupported ORDER BY key type");
                return (int)kerr;
            }
            uint64_t mmin = h_bins[2049], mmax = h_bins[2050];
            if (tp.level0) {
                null_cnt = h_bins[2048];
                nonnull = 0;
                for (int i = 0; i < 2048; i++) nonnull += (int64_t)h_bins[i];
                const bool nulls_first = !desc0;
                int64_t null_take = nulls_first
                    ? ((int64_t)null_cnt < K ? (int64_t)null_cnt : K)
                    : (K - nonnull > 0 ? K - nonnull : 0);
                k_nonnull = K - (nulls_first ? null_take : 0);
                if (!nulls_first && k_nonnull > nonnull) k_nonnull = nonnull;
                if (k_nonnull <= 0) { only_nulls = true; break; }
                if (k_nonnull >= nonnull) { all_nonnull = true; S = nonnull; break; }
            }
            int64_t cum = 0;
            int t = -1;
            for (int i = 0; i < 2048; i++) {
                if (S + cum + (int64_t)h_bins[i] >= k_nonnull) { t = i; break; }
                cum += (int64_t)h_bins[i];
            }
            if (t < 0) {   /* cannot happen: counts shrank between passes */
                g_pool.put(d_bins);
                set_err(errbuf, errlen, "ORDER BY selection internal error");
                return YT_ERR_HIP;
            }
            S += cum;
            T = (int64_t)h_bins[t];
            /* shrink [lo, hi] to the chosen bin (∩ data bounds on pass 0) */
            uint64_t nlo = lo + ((uint64_t)t << shift);
            uint64_t span = (uint64_t)(t + 1) << shift;
            uint64_t nhi = (span == 0 || nlo + (span - ((uint64_t)t << shift)) - 1 < nlo)
                ? hi : nlo + (((uint64_t)1 << shift) - 1);
            if (nhi > hi) nhi = hi;
            if (tp.level0) {
                if (mmin > nlo) nlo = mmin;
                if (mmax < nhi) nhi = mmax;
            }
            lo = nlo;
            hi = nhi;
            if (T <= CAPB || shift == 0 || lo >= hi) break;
            uint64_t range = hi - lo;     /* >= 1 */
            shift = 0;
            while ((range >> shift) >= 2048) shift++;
            tp.lo = lo;
            tp.hi = hi;
            tp.shift = shift;
            tp.level0 = 0;
        }

        if (!all_nonnull && !only_nulls && T > CAPB && plan->order_count > 1) {
            g_pool.put(d_bins);
            set_err(errbuf, errlen,
                    "order-key tie set too large for multi-key ORDER BY this round");
            return YT_ERR_UNSUPPORTED;
        }
        const int64_t CAPN = K + CAPB;
        {
            const bool nulls_first = !desc0;
            int64_t null_take = nulls_first
                ? ((int64_t)null_cnt < K ? (int64_t)null_cnt : K)
                : (K - nonnull > 0 ? K - nonnull : 0);
            if ((int64_t)null_cnt > CAPN && plan->order_count > 1 &&
                null_take > 0 && null_take < (int64_t)null_cnt) {
                g_pool.put(d_bins);
                set_err(errbuf, errlen,
                        "null order-key set too large for multi-key ORDER BY this round");
                return YT_ERR_UNSUPPORTED;
            }
        }

        /* gather candidates: strict (exactly S), ties (cap), nulls (cap) */
        TopkGather tg;
        tg.order_proj = ord0;
        tg.desc = desc0;
        tg.all_nonnull = all_nonnull ? 1 : 0;
        tg.pad_ = 0;
        /* only_nulls: nothing strict (strict is m < lo, so lo = 0 matches
         * no row — a mapped key CAN be 0, e.g. INT64_MIN ascending, and the
         * strict buffer is unbounded) and at most the capped tie bucket */
        tg.lo = only_nulls ? 0 : lo;
        tg.hi = only_nulls ? 0 : hi;
        tg.cap_tie = CAPB;
        tg.cap_null = CAPN;
        int64_t* d_rows_strict = nullptr;
        int64_t* d_rows_tie = nullptr;
        int64_t* d_rows_null = nullptr;
        HIP_CHECK(pool_alloc(&d_rows_strict, sizeof(int64_t) * (S ? S : 1)));
        HIP_CHECK(pool_alloc(&d_rows_tie, sizeof(int64_t) * CAPB));
        HIP_CHECK(pool_alloc(&d_rows_null, sizeof(int64_t) * CAPN));
        HIP_CHECK(hipMemsetAsync(d_ctrs, 0, 3 * sizeof(unsigned long long),
                                 R2.stream));
        if (fast_col >= 0) {
            HIP_CHECK(ytql_launch_topk_gather_fast(
                R2.d_segs, R2.d_segex, R2.h_off[fast_col], fast_shift, n,
                fast_nulls, fast_signed, &tg,
                d_rows_strict, d_ctrs, d_rows_tie, d_ctrs + 1,
                d_rows_null, d_ctrs + 2, R2.stream));
        } else {
            HIP_CHECK(ytql_launch_topk_gather(dp, R2.d_segs, R2.d_segex,
                                              R2.d_off, R2.d_cnt, n, jd,
                                              jd2, &tg,
                                              d_rows_strict, d_ctrs,
                                              d_rows_tie, d_ctrs + 1,
                                              d_rows_null, d_ctrs + 2,
                                              R2.d_err, R2.stream));
        }
        launches++;
        unsigned long long hctrs[3];
        HIP_CHECK(hipMemcpy(hctrs, d_ctrs, 3 * sizeof(unsigned long long),
                            hipMemcpyDeviceToHost));
        int64_t nS = (int64_t)hctrs[0];
        int64_t nB = (int64_t)hctrs[1] < CAPB ? (int64_t)hctrs[1] : CAPB;
        int64_t nN = (int64_t)hctrs[2] < CAPN ? (int64_t)hctrs[2] : CAPN;
        int64_t M = nS + nB + nN;

        /* pack the three lists and materialize the projected rows */
        int64_t* d_rows_all = nullptr;
        DevOutVal* d_vals = nullptr;
        DevOutVal* h_vals = nullptr;
        HIP_CHECK(pool_alloc(&d_rows_all, sizeof(int64_t) * (M ? M : 1)));
        if (nS) HIP_CHECK(hipMemcpyAsync(d_rows_all, d_rows_strict,
                                         sizeof(int64_t) * nS,
                                         hipMemcpyDeviceToDevice, R2.stream));
        if (nB) HIP_CHECK(hipMemcpyAsync(d_rows_all + nS, d_rows_tie,
                                         sizeof(int64_t) * nB,
                                         hipMemcpyDeviceToDevice, R2.stream));
        if (nN) HIP_CHECK(hipMemcpyAsync(d_rows_all + nS + nB, d_rows_null,
                                         sizeof(int64_t) * nN,
                                         hipMemcpyDeviceToDevice, R2.stream));
        HIP_CHECK(pool_alloc(&d_vals, sizeof(DevOutVal) * (M ? M : 1) * np));
        HIP_CHECK(pool_alloc_host(&h_vals, sizeof(DevOutVal) * (M ? M : 1) * np));
        if (M) {
            HIP_CHECK(ytql_launch_topk_materialize(dp, R2.d_segs, R2.d_segex,
                                                   R2.d_off, R2.d_cnt, jd, jd2,
                                                   d_rows_all, M, d_vals,
                                                   R2.d_err, R2.stream));
            launches++;
            HIP_CHECK(hipMemcpyAsync(h_vals, d_vals, sizeof(DevOutVal) * M * np,
                                     hipMemcpyDeviceToHost, R2.stream));
        }
        HIP_CHECK(hipEventRecord(e1, R2.stream));
        HIP_CHECK(hipStreamSynchronize(R2.stream));
        if (stats) stats->kernel_other_ms = now_ms() - tw0;  /* pre-emit wall */
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
        (void)hipEventDestroy(e0);
        (void)hipEventDestroy(e1);
        unsigned kerr = 0;
        HIP_CHECK(hipMemcpy(&kerr, R2.d_err, sizeof(unsigned),
                            hipMemcpyDeviceToHost));
        auto put_all = [&]() {
            g_pool.put(d_bins); g_pool.put(d_rows_strict); g_pool.put(d_rows_tie);
            g_pool.put(d_rows_null); g_pool.put(d_rows_all); g_pool.put(d_vals);
            g_pool.put(h_vals);
        };
        if (kerr) {
            put_all();
            if (kerr == 100) {
                set_err(errbuf, errlen, "NaN in ORDER BY comparison");
                return YT_ERR_LIMIT;
            }
            set_err(errbuf, errlen,
                    kerr == YT_ERR_DIV_ZERO ? "Division by zero" : "expression error");
            return (int)kerr;
        }

        /* host: full-comparer sort of the candidates, emit the slice */
        std::vector<int64_t> idx((size_t)M);
        for (int64_t i = 0; i < M; i++) idx[i] = i;
        int nan_err = 0;
        auto cmp_cand = [&](int64_t ia, int64_t ib) {
            const DevOutVal* a = h_vals + ia * np;
            const DevOutVal* b = h_vals + ib * np;
            for (int i = 0; i < plan->order_count; i++) {
                int c = plan->order_cols[i];
                const DevOutVal& x = a[c];
                const DevOutVal& y = b[c];
                int xn = x.type == YT_VT_NULL, yn = y.type == YT_VT_NULL;
                int r = 0;
                if (xn || yn) {
                    r = (xn == yn) ? 0 : (xn ? -1 : 1);
                } else if (x.type == YT_VT_DOUBLE) {
                    double xv, yv;
                    memcpy(&xv, &x.bits, 8);
                    memcpy(&yv, &y.bits, 8);
                    if (xv != xv || yv != yv) { nan_err = 1; return false; }
                    r = xv < yv ? -1 : (xv > yv ? 1 : 0);
                } else if (x.type == YT_VT_INT64) {
                    int64_t xv = (int64_t)x.bits, yv = (int64_t)y.bits;
                    r = xv < yv ? -1 : (xv > yv ? 1 : 0);
                } else {
                    r = x.bits < y.bits ? -1 : (x.bits > y.bits ? 1 : 0);
                }
                if (plan->order_desc && plan->order_desc[i]) r = -r;
                if (r) return r < 0;
            }
            return false;
        };
        std::sort(idx.begin(), idx.end(), cmp_cand);
        if (nan_err) {
            put_all();
            set_err(errbuf, errlen, "NaN in ORDER BY comparison");
            return YT_ERR_LIMIT;
        }
        int64_t b = plan->order_offset < M ? plan->order_offset : M;
        int64_t e = b + plan->order_limit;
        if (e > M) e = M;
        int out_limited = 0;
        for (int64_t i = b; i < e; i++) {
            if (options->output_row_limit > 0 &&
                output->row_count >= options->output_row_limit) {
                out_limited = 1;
                break;
            }
            if (output->row_count >= output->capacity_rows) {
                put_all();
                return YT_ERR_CAPACITY;
            }
            const DevOutVal* src = h_vals + idx[i] * np;
            YtValue* dst = output->values + output->row_count * np;
            for (int pj = 0; pj < np; pj++) {
                dst[pj].id = (uint16_t)pj;
                dst[pj].type = (uint8_t)src[pj].type;
                dst[pj].flags = 0;
                dst[pj].length = 0;
                dst[pj].data.bits = src[pj].bits;
            }
            output->row_count++;
        }
        if (stats) {
            stats->rows_read = n;
            stats->rows_written = output->row_count;
            stats->incomplete_output = out_limited;
            stats->kernel_scan_ms += ms;
            stats->kernel_scan_launches += launches;
            stats->execute_time_ms = now_ms() - tw0;
        }
        put_all();
    }
    return YT_OK;
fail:
    return rc;
}

/* scan + filter + project: order-preserving (MakeCodegenProjectOp +
 * WriteOpHelper, registry.cpp:1999-2047). The device writes per-row values
 * and a pass mask; the host compacts in row order. */
static int run_scan_project(const YtPlan* plan, const YtChunk* chunk,
                            const YtExecOptions* options, const DevPlan* dp,
                            const JoinDev* jd, const JoinDev* jd2,
                            YtRowset* output, YtStatistics* stats, double tw0,
                            char* errbuf, size_t errlen)
{
    int rc = YT_OK;
    DeviceRun R2;
    R2.stream = (hipStream_t)(uintptr_t)options->stream;
    unsigned mw = 0;
    int in_clamped = 0;
    rc = setup_chunk(chunk, &R2, &mw, options->input_row_limit, &in_clamped,
                     errbuf, errlen);
    if (rc) return rc;
    (void)in_clamped;
    int64_t n = chunk->row_count;
    if (options->input_row_limit > 0 && options->input_row_limit < n)
        n = options->input_row_limit;
    output->row_count = 0;
    output->column_count = plan->project_count;
    if (n == 0 || R2.nsegs == 0) return YT_OK;
    {
        /* stream the scan through bounded windows: the per-row
         * materialization buffer is at most kWin rows, so arbitrarily
         * large inputs need only bounded device/host memory */
        const int64_t kWin = (int64_t)1 << 24;
        const int64_t win = n < kWin ? n : kWin;
        DevOutVal* d_out = nullptr;
        uint8_t* d_pass = nullptr;
        DevOutVal* h_out = nullptr;
        uint8_t* h_pass = nullptr;
        HIP_CHECK(pool_alloc(&d_out, sizeof(DevOutVal) * win * plan->project_count));
        HIP_CHECK(pool_alloc(&d_pass, (size_t)win));
        HIP_CHECK(pool_alloc_host(&h_out, sizeof(DevOutVal) * win * plan->project_count));
        HIP_CHECK(pool_alloc_host(&h_pass, (size_t)win));
        HIP_CHECK(hipMemsetAsync(R2.d_err, 0, sizeof(unsigned), R2.stream));
        float ms = 0;
        int np = plan->project_count;
        for (int64_t w0 = 0; w0 < n; w0 += win) {
        const int64_t wlen = (w0 + win <= n) ? win : (n - w0);
        hipEvent_t e0, e1;
        HIP_CHECK(hipEventCreate(&e0));
        HIP_CHECK(hipEventCreate(&e1));
        HIP_CHECK(hipEventRecord(e0, R2.stream));
        HIP_CHECK(ytql_launch_scan_project(dp, R2.d_segs, R2.d_segex, R2.d_off,
                                           R2.d_cnt, w0, wlen, jd, jd2, d_out,
                                           d_pass, R2.d_err, R2.stream));
        HIP_CHECK(hipEventRecord(e1, R2.stream));
        HIP_CHECK(hipMemcpyAsync(h_out, d_out, sizeof(DevOutVal) * wlen * plan->project_count,
                                 hipMemcpyDeviceToHost, R2.stream));
        HIP_CHECK(hipMemcpyAsync(h_pass, d_pass, (size_t)wlen,
                                 hipMemcpyDeviceToHost, R2.stream));
        HIP_CHECK(hipStreamSynchronize(R2.stream));
        float wms = 0;
        HIP_CHECK(hipEventElapsedTime(&wms, e0, e1));
        ms += wms;
        (void)hipEventDestroy(e0);
        (void)hipEventDestroy(e1);
        unsigned kerr = 0;
        HIP_CHECK(hipMemcpy(&kerr, R2.d_err, sizeof(unsigned), hipMemcpyDeviceToHost));
        if (kerr) {
            g_pool.put(d_out); g_pool.put(d_pass);
            g_pool.put(h_out); g_pool.put(h_pass);
            set_err(errbuf, errlen, kerr == YT_ERR_DIV_ZERO ? "Division by zero" : "expression error");
            return (int)kerr;
        }
        for (int64_t r2 = 0; r2 < wlen; r2++) {
            if (!h_pass[r2]) continue;
            if (options->output_row_limit > 0 &&
                output->row_count >= options->output_row_limit) {
                if (stats) stats->incomplete_output = 1;
                break;
            }
            if (output->row_count >= output->capacity_rows) {
                g_pool.put(d_out); g_pool.put(d_pass);
                g_pool.put(h_out); g_pool.put(h_pass);
                return YT_ERR_CAPACITY;
            }
            YtValue* dst = output->values + output->row_count * np;
            for (int pj = 0; pj < np; pj++) {
                const DevOutVal& o = h_out[r2 * np + pj];
                dst[pj].id = (uint16_t)pj;
                dst[pj].type = (uint8_t)o.type;
                dst[pj].flags = 0;
                dst[pj].length = 0;
                dst[pj].data.bits = o.bits;
            }
            output->row_count++;
        }
        if (options->output_row_limit > 0 &&
            output->row_count >= options->output_row_limit)
            break;
        }
        if (stats) {
            stats->rows_read = n;
            stats->rows_written = output->row_count;
            stats->kernel_scan_ms += ms;
            stats->kernel_scan_launches += 1;
            stats->execute_time_ms = now_ms() - tw0;
        }
        g_pool.put(d_out); g_pool.put(d_pass);
        g_pool.put(h_out); g_pool.put(h_pass);
    }
    return YT_OK;
fail:
    return rc;
}


/* string-keyed GROUP BY (config-5 family): dictionary-encoded string key,
 * aggregates {sum(col), sum(1)}. Dense per-segment id accumulators, then an
 * exact cross-segment merge (hash gate + byte compare), then device-side
 * key materialization into the caller's string pool. */
static int run_string_group(const YtPlan* plan, const YtChunk* chunk,
                            const YtExecOptions* options, int key_col,
                            YtRowset* output, YtStatistics* stats, double tw0,
                            char* errbuf, size_t errlen,
                            StrPartialOut* po = nullptr)
{
    int rc = YT_OK;
    int sum_slot = -1, sum_col = -1, val_is_double = 0;
    for (int a = 0; a < plan->agg_count; a++) {
        int f = plan->aggs[a]->func;
        if (f == YT_AGG_SUM) {
            int c;
            if (sum_slot >= 0 || !expr_is_col(plan->aggs[a]->arg, &c)) {
                set_err(errbuf, errlen, "string keys: one direct sum this round");
                return YT_ERR_UNSUPPORTED;
            }
            sum_slot = a;
            sum_col = c;
            val_is_double = chunk->columns[c].value_type == YT_VT_DOUBLE;
            if (!val_is_double && chunk->columns[c].value_type != YT_VT_INT64) {
                set_err(errbuf, errlen, "string keys: int64/double sums this round");
                return YT_ERR_UNSUPPORTED;
            }
        } else if (f != YT_AGG_SUM1) {
            set_err(errbuf, errlen, "string keys: sum/sum(1) this round");
            return YT_ERR_UNSUPPORTED;
        }
    }
    if (plan->project_count) {
        set_err(errbuf, errlen, "string keys: no post-projection this round");
        return YT_ERR_UNSUPPORTED;
    }

    DeviceRun R;
    R.stream = (hipStream_t)(uintptr_t)options->stream;
    unsigned mw = 0;
    int in_clamped = 0;
    rc = setup_chunk(chunk, &R, &mw, options->input_row_limit, &in_clamped,
                     errbuf, errlen);
    if (rc) return rc;
    if (output) {
        output->row_count = 0;
        output->string_pool_used = 0;   /* rowsets reusable across queries */
        output->column_count = 1 + plan->agg_count;
    }
    if (chunk->row_count == 0 || R.nsegs == 0) {
        if (po) {
            for (int p = 0; p < po->partition_count; p++) {
                po->part_counts[p] = 0;
                po->part_pool_bytes[p] = 0;
            }
        }
        return YT_OK;
    }
    rc = setup_table(&R, plan->agg_count, options->max_groups_hint,
                     options->group_row_limit, errbuf, errlen);
    if (rc) return rc;

    /* read back the parsed key-column segments: dictionary sizes + layout */
    int knseg = R.h_cnt[key_col];
    int koff = R.h_off[key_col];
    std::vector<SegEx> ksegex(knseg);
    TableHdr th;
    int64_t total_dict = 0;
    {
        HIP_CHECK(hipMemcpy(ksegex.data(), R.d_segex + koff,
                            sizeof(SegEx) * knseg, hipMemcpyDeviceToHost));
        for (int i = 0; i < knseg; i++) {
            if (!(ksegex[i].flags & (8 | 32))) {
                set_err(errbuf, errlen,
                        "string keys: dictionary or direct-dense segments "
                        "(RLE strings not this round)");
                return YT_ERR_UNSUPPORTED;
            }
        }
        int32_t rows0 = R.h_segs[koff].row_count;
        for (int i = 0; i + 1 < knseg; i++) {
            if (R.h_segs[koff + i].row_count != rows0) {
                set_err(errbuf, errlen, "string keys: non-uniform segments");
                return YT_ERR_UNSUPPORTED;
            }
        }
        if (sum_col >= 0 && R.h_cnt[sum_col] != knseg) {
            set_err(errbuf, errlen, "string keys: column segmentation mismatch");
            return YT_ERR_UNSUPPORTED;
        }

        std::vector<int64_t> acc_base(knseg + 1);
        for (int i = 0; i < knseg; i++) {
            acc_base[i] = total_dict;
            total_dict += ksegex[i].dict_size;
        }
        acc_base[knseg] = total_dict;

        StrGroupParams sp;
        memset(&sp, 0, sizeof(sp));
        sp.key_seg_off = koff;
        sp.key_seg_cnt = knseg;
        sp.val_seg_off = sum_col >= 0 ? R.h_off[sum_col] : -1;
        sp.val_seg_cnt = sum_col >= 0 ? R.h_cnt[sum_col] : 0;
        sp.val_is_double = val_is_double;
        sp.sum_slot = sum_slot;
        sp.agg_count = plan->agg_count;
        sp.has_val_nulls = sum_col >= 0 && (R.col_null_flags[sum_col] != 0);
        sp.tile_rows = 8192;
        if (sp.tile_rows > rows0) sp.tile_rows = rows0 > 256 ? rows0 : 256;
        sp.tiles_per_seg = (rows0 + sp.tile_rows - 1) / sp.tile_rows;
        {
            const char* sa = getenv("YTQL_STRACCUM");
            sp.xcd_affine = sa ? atoi(sa) : 1;
        }
        int32_t last_rows = R.h_segs[koff + knseg - 1].row_count;
        sp.ntiles = (knseg - 1) * sp.tiles_per_seg
                  + (last_rows + sp.tile_rows - 1) / sp.tile_rows;
        sp.row_count = chunk->row_count;

        int64_t* d_accbase = nullptr;
        unsigned long long* d_acc = nullptr;
        uint64_t* d_hashes = nullptr;
        StrSlot* d_slots = nullptr;
        HIP_CHECK(pool_alloc(&d_accbase, sizeof(int64_t) * (knseg + 1)));
        HIP_CHECK(hipMemcpyAsync(d_accbase, acc_base.data(),
                                 sizeof(int64_t) * (knseg + 1),
                                 hipMemcpyHostToDevice, R.stream));
        HIP_CHECK(pool_alloc(&d_acc, sizeof(uint64_t) * 2 * (total_dict ? total_dict : 1)));
        HIP_CHECK(hipMemsetAsync(d_acc, 0, sizeof(uint64_t) * 2 * (total_dict ? total_dict : 1), R.stream));
        HIP_CHECK(pool_alloc(&d_hashes, sizeof(uint64_t) * (total_dict ? total_dict : 1)));
        /* per-entry identity + prefix arrays: filled by the hash kernel so
         * the merge's hot path reads them as a stream (YTQL_STRMERGE=0
         * falls back to the round-1 dict-read probe for A/B) */
        uint64_t* d_idents = nullptr;
        ulonglong2* d_pfxs = nullptr;
        HIP_CHECK(pool_alloc(&d_idents, sizeof(uint64_t) * (total_dict ? total_dict : 1)));
        HIP_CHECK(pool_alloc(&d_pfxs, sizeof(ulonglong2) * (total_dict ? total_dict : 1)));
        const char* sm_env = getenv("YTQL_STRMERGE");
        int merge_fast = sm_env ? atoi(sm_env) : 1;
        /* Table sized for the worst case (every dict entry distinct).
         * Measured: an 8×-smaller table holding just the ~6 M live keys is
         * NOT faster — the merge's scattered atomics then fight over hot
         * cachelines across XCDs (59 ms vs 47 ms at 86 M entries), while
         * the oversized cold table costs only compact-scan time, which the
         * two-pass compact already made cheap. The grow-on-overflow loop
         * below is kept as a safety net (never triggers at this size). */
        uint64_t nslots_cap = next_pow2((uint64_t)(total_dict ? total_dict : 1) * 2);
        if (nslots_cap < 2048) nslots_cap = 2048;
        /* Start smaller than the every-entry-distinct worst case: the 64 B/slot
         * table is memset before the merge and scanned by the compact, so an
         * oversized table costs real milliseconds per step. With a caller
         * hint, 2x the hint; otherwise half the entry count (cross-segment
         * duplication >= 2x in any chunk where a GROUP BY is the right
         * plan). The grow-on-overflow loop below keeps correctness for
         * adversarial cardinalities (one 4x retry reaches 2x entries). */
        const char* sx_env = getenv("YTQL_STRSLOTS_X");
        uint64_t slots_x = sx_env ? (uint64_t)atoll(sx_env) : 4;
        if (slots_x < 1) slots_x = 1;
        uint64_t nslots = options->max_groups_hint > 0
            ? next_pow2((uint64_t)options->max_groups_hint * slots_x)
            : next_pow2((uint64_t)(total_dict ? total_dict : 1) / 2 + 1);
        if (nslots < 2048) nslots = 2048;
        if (nslots > nslots_cap) nslots = nslots_cap;
        HIP_CHECK(pool_alloc(&d_slots, sizeof(StrSlot) * nslots));
        HIP_CHECK(hipMemsetAsync(d_slots, 0, sizeof(StrSlot) * nslots, R.stream));

        hipEvent_t e0, e1, e2, ea, eh;
        HIP_CHECK(hipEventCreate(&e0));
        HIP_CHECK(hipEventCreate(&e1));
        HIP_CHECK(hipEventCreate(&e2));
        HIP_CHECK(hipEventCreate(&ea));
        HIP_CHECK(hipEventCreate(&eh));
        HIP_CHECK(hipEventRecord(e0, R.stream));
        HIP_CHECK(ytql_launch_strgrp_accum(&sp, R.d_segs, R.d_segex, d_accbase,
                                           d_acc, R.d_th, R.stream));
        HIP_CHECK(hipEventRecord(ea, R.stream));
        HIP_CHECK(ytql_launch_strgrp_hash(R.d_segs, R.d_segex, koff, knseg,
                                          d_accbase, d_hashes, d_idents, d_pfxs,
                                          total_dict, R.stream));
        HIP_CHECK(hipEventRecord(eh, R.stream));
        for (;;) {
            HIP_CHECK(ytql_launch_strgrp_merge(R.d_segs, R.d_segex, koff, knseg,
                                               d_accbase, d_acc, d_hashes,
                                               d_idents, d_pfxs,
                                               d_slots, nslots, val_is_double,
                                               merge_fast, R.d_th, total_dict,
                                               R.stream));
            HIP_CHECK(hipMemcpy(&th, R.d_th, sizeof(th), hipMemcpyDeviceToHost));
            if (th.overflow != 1 || nslots >= nslots_cap) break;
            g_pool.put(d_slots);
            nslots *= 4;
            if (nslots > nslots_cap) nslots = nslots_cap;
            HIP_CHECK(pool_alloc(&d_slots, sizeof(StrSlot) * nslots));
            HIP_CHECK(hipMemsetAsync(d_slots, 0, sizeof(StrSlot) * nslots, R.stream));
            th.ngroups = 0;
            th.overflow = 0;
            HIP_CHECK(hipMemcpy(R.d_th, &th, sizeof(th), hipMemcpyHostToDevice));
        }
        HIP_CHECK(hipEventRecord(e1, R.stream));

        /* compact + materialize keys into a device pool */
        int64_t ngroups = (int64_t)th.ngroups;
        uint64_t pool_cap = 0;
        for (int i = 0; i < knseg; i++) {
            /* dictionary blob bytes upper-bound the key bytes */
            pool_cap += (uint64_t)R.h_segs[koff + i].blob_bytes;
        }
        if (pool_cap < 1024) pool_cap = 1024;
        OutStrGroup* d_out = nullptr;
        char* d_pool = nullptr;
        unsigned long long* d_ctr = nullptr;
        HIP_CHECK(pool_alloc(&d_out, sizeof(OutStrGroup) * (ngroups ? ngroups : 1)));
        HIP_CHECK(pool_alloc(&d_pool, pool_cap));
        HIP_CHECK(pool_alloc(&d_ctr, 2 * sizeof(unsigned long long)));
        HIP_CHECK(hipMemsetAsync(d_ctr, 0, 2 * sizeof(unsigned long long), R.stream));
        if (ngroups > 0) {
            HIP_CHECK(ytql_launch_strgrp_compact(R.d_segs, R.d_segex, koff,
                                                 d_slots, nslots, d_out, d_ctr,
                                                 d_pool, d_ctr + 1, pool_cap,
                                                 R.d_th, R.stream));
        }
        OutStrGroup* hgroups = nullptr;
        HIP_CHECK(pool_alloc_host(&hgroups,
                                  sizeof(OutStrGroup) * (ngroups ? ngroups : 1)));
        unsigned long long hctr[2] = {0, 0};
        /* the fast emit path below streams hgroups in SLICES so the host
         * emit overlaps the remaining D2H; the limited path copies here */
        HIP_CHECK(hipMemcpyAsync(hctr, d_ctr, 2 * sizeof(unsigned long long),
                                 hipMemcpyDeviceToHost, R.stream));
        HIP_CHECK(hipEventRecord(e2, R.stream));
        HIP_CHECK(hipStreamSynchronize(R.stream));
        float ms = 0, ms_other = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
        HIP_CHECK(hipEventElapsedTime(&ms_other, e1, e2));
        if (getenv("YTQL_TIMING")) {
            float ms_acc = 0, ms_hash = 0;
            HIP_CHECK(hipEventElapsedTime(&ms_acc, e0, ea));
            HIP_CHECK(hipEventElapsedTime(&ms_hash, ea, eh));
            fprintf(stderr, "[ytql timing] strgrp: accum %.2fms hash %.2fms "
                    "merge %.2fms compact+D2H %.2fms\n",
                    ms_acc, ms_hash, ms - ms_acc - ms_hash, ms_other);
        }
        (void)hipEventDestroy(ea);
        (void)hipEventDestroy(eh);
        (void)hipEventDestroy(e0);
        (void)hipEventDestroy(e1);
        (void)hipEventDestroy(e2);
        HIP_CHECK(hipMemcpy(&th, R.d_th, sizeof(th), hipMemcpyDeviceToHost));
        if (th.overflow == 1) {
            g_pool.put(d_accbase); g_pool.put(d_acc); g_pool.put(d_hashes);
            g_pool.put(d_idents); g_pool.put(d_pfxs);
            g_pool.put(d_slots); g_pool.put(d_out); g_pool.put(d_pool);
            g_pool.put(d_ctr); g_pool.put(hgroups);
            set_err(errbuf, errlen, "string merge table/pool overflow");
            return YT_ERR_CAPACITY;
        }

        if (po) {
            /* bottom query: hash-partition the compacted groups + the null
             * side group into state rows and per-partition pool slices
             * (key_bits = slice-local offset<<24 | len) */
            auto putall = [&]() {
                g_pool.put(d_accbase); g_pool.put(d_acc); g_pool.put(d_hashes);
                g_pool.put(d_idents); g_pool.put(d_pfxs);
                g_pool.put(d_slots); g_pool.put(d_out); g_pool.put(d_pool);
                g_pool.put(d_ctr); g_pool.put(hgroups);
            };
            const int np = po->partition_count;
            const int p_null = (int)(splitmix64_host(
                0xCBF29CE484222325ULL ^ 0xDEADBEEF12345678ULL) % (uint64_t)np);
            unsigned long long* d_pc = nullptr;   /* [np] counts, [np] bytes,
                                                     [np] cursors, [np] bcursors,
                                                     [np] pool_base */
            HIP_CHECK(pool_alloc(&d_pc, sizeof(unsigned long long) * np * 5));
            HIP_CHECK(hipMemsetAsync(d_pc, 0,
                                     sizeof(unsigned long long) * np * 5,
                                     R.stream));
            if (ngroups > 0) {
                HIP_CHECK(ytql_launch_strst_count(d_out, ngroups, d_pool, np,
                                                  d_pc, d_pc + np, R.stream));
            }
            std::vector<unsigned long long> pc(2 * np);
            HIP_CHECK(hipMemcpyAsync(pc.data(), d_pc,
                                     sizeof(unsigned long long) * 2 * np,
                                     hipMemcpyDeviceToHost, R.stream));
            HIP_CHECK(hipStreamSynchronize(R.stream));
            if (th.side_used[1]) pc[p_null] += 1;
            unsigned long long rrun = 0, brun = 0;
            std::vector<unsigned long long> rcur(np), pbase(np);
            for (int p = 0; p < np; p++) {
                rcur[p] = rrun;
                pbase[p] = brun;
                rrun += pc[p];
                brun += pc[np + p];
            }
            if ((int64_t)rrun > po->capacity_rows
                || (int64_t)brun > po->pool_capacity) {
                putall();
                g_pool.put(d_pc);
                set_err(errbuf, errlen, "partial_str: state/pool buffer too small");
                return YT_ERR_CAPACITY;
            }
            HIP_CHECK(hipMemcpyAsync(d_pc + 2 * np, rcur.data(),
                                     sizeof(unsigned long long) * np,
                                     hipMemcpyHostToDevice, R.stream));
            HIP_CHECK(hipMemsetAsync(d_pc + 3 * np, 0,
                                     sizeof(unsigned long long) * np, R.stream));
            HIP_CHECK(hipMemcpyAsync(d_pc + 4 * np, pbase.data(),
                                     sizeof(unsigned long long) * np,
                                     hipMemcpyHostToDevice, R.stream));
            if (ngroups > 0) {
                HIP_CHECK(ytql_launch_strst_scatter(d_out, ngroups, d_pool, np,
                                                    val_is_double,
                                                    d_pc + 2 * np, d_pc + 3 * np,
                                                    po->states_device,
                                                    po->pool_device,
                                                    d_pc + 4 * np, R.stream));
            }
            HIP_CHECK(hipStreamSynchronize(R.stream));
            if (th.side_used[1]) {
                unsigned long long pos = 0;
                HIP_CHECK(hipMemcpy(&pos, d_pc + 2 * np + p_null,
                                    sizeof(pos), hipMemcpyDeviceToHost));
                YtStateRow sr;
                memset(&sr, 0, sizeof(sr));
                sr.key_bits = 0;
                uint64_t nn = sum_slot >= 0 ? th.side_agg[1][2 * sum_slot + 1] : 0;
                sr.meta = 1ULL | (val_is_double ? 2ULL : 0ULL) | (nn << 8);
                sr.sum_bits = sum_slot >= 0 ? th.side_agg[1][2 * sum_slot] : 0;
                sr.row_count = th.side_cnt[1];
                HIP_CHECK(hipMemcpy(po->states_device + pos, &sr, sizeof(sr),
                                    hipMemcpyHostToDevice));
            }
            for (int p = 0; p < np; p++) {
                po->part_counts[p] = (int64_t)pc[p];
                po->part_pool_bytes[p] = (int64_t)pc[np + p];
            }
            if (stats) {
                stats->rows_read = chunk->row_count;
                stats->grouped_row_count = (int64_t)rrun;
                stats->incomplete_input = in_clamped;
                stats->incomplete_output = (th.overflow == 2);
                stats->kernel_scan_ms += ms;
                stats->kernel_scan_launches += 1;
                stats->kernel_other_ms += ms_other;
                stats->execute_time_ms = now_ms() - tw0;
            }
            putall();
            g_pool.put(d_pc);
            return YT_OK;
        }

        /* emit [key(string), aggs...] rows. Fast path (no output limit,
         * everything fits): copy the device key pool wholesale — compact's
         * pool_off values are global offsets into it — and fill the YtValue
         * rows with host threads. */
        int ncols = 1 + plan->agg_count;
        int out_limited = 0;
        int64_t ng = ngroups;
        int has_null_row = th.side_used[1] ? 1 : 0;
        if (options->output_row_limit == 0 &&
            ng + has_null_row <= output->capacity_rows &&
            (unsigned long long)hctr[1] <= (unsigned long long)output->string_pool_capacity &&
            (output->string_pool != nullptr || hctr[1] == 0)) {
            /* pipeline: pool D2H first, then hgroups in slices — the
             * parallel host copies/emits of slice s overlap slice s+1's
             * D2H (the serial tail was D2H then copy then emit) */
            char* stage = nullptr;
            if (hctr[1]) {
                HIP_CHECK(pool_alloc_host(&stage, hctr[1]));
                HIP_CHECK(hipMemcpyAsync(stage, d_pool, hctr[1],
                                         hipMemcpyDeviceToHost, R.stream));
            }
            const int kSlices = (ng >= (1 << 21)) ? 8 : 1;
            std::vector<hipEvent_t> sl_ev(kSlices, nullptr);
            int64_t per_sl = (ng + kSlices - 1) / kSlices;
            for (int s2 = 0; s2 < kSlices; s2++) {
                int64_t b = (int64_t)s2 * per_sl;
                int64_t e = std::min<int64_t>(ng, b + per_sl);
                if (b >= e) break;
                HIP_CHECK(hipMemcpyAsync(hgroups + b, d_out + b,
                                         sizeof(OutStrGroup) * (e - b),
                                         hipMemcpyDeviceToHost, R.stream));
                HIP_CHECK(hipEventCreate(&sl_ev[s2]));
                HIP_CHECK(hipEventRecord(sl_ev[s2], R.stream));
            }
            std::thread pool_thr;
            if (hctr[1]) {
                /* pool D2H was queued first — wait for the FIRST slice event
                 * (>= pool completion), then copy to the caller's pool off
                 * the critical path */
                hipEvent_t after_pool = sl_ev[0];
                if (!after_pool) {
                    HIP_CHECK(hipStreamSynchronize(R.stream));
                }
                size_t total_b = (size_t)hctr[1];
                char* dst_pool = output->string_pool;
                pool_thr = std::thread([stage, dst_pool, total_b, after_pool] {
                    if (after_pool) (void)hipEventSynchronize(after_pool);
                    int ct = (int)std::min<size_t>(
                        std::thread::hardware_concurrency(),
                        (total_b + (64 << 20) - 1) / (64 << 20));
                    if (ct > 1) {
                        std::vector<std::thread> cth;
                        size_t per = (total_b + ct - 1) / ct;
                        for (int t = 0; t < ct; t++) {
                            size_t b = (size_t)t * per;
                            size_t e = std::min(total_b, b + per);
                            if (b >= e) break;
                            cth.emplace_back([=] {
                                memcpy(dst_pool + b, stage + b, e - b);
                            });
                        }
                        for (auto& th2 : cth) th2.join();
                    } else if (total_b) {
                        memcpy(dst_pool, stage, total_b);
                    }
                });
            }
            output->string_pool_used = hctr[1];
            int agg_is_sum1[kMaxAggs];
            for (int a = 0; a < plan->agg_count; a++)
                agg_is_sum1[a] = plan->aggs[a]->func == YT_AGG_SUM1;
            auto emit_range = [&](int64_t b, int64_t e) {
                for (int64_t gI = b; gI < e; gI++) {
                    const OutStrGroup& g = hgroups[gI];
                    YtValue* dst = output->values + gI * ncols;
                    dst[0].id = 0;
                    dst[0].flags = 0;
                    dst[0].type = YT_VT_STRING;
                    dst[0].length = (uint32_t)(g.off_len & 0xFFFFFF);
                    dst[0].data.str = output->string_pool + (g.off_len >> 24);
                    for (int a = 0; a < plan->agg_count; a++) {
                        YtValue& v = dst[1 + a];
                        v.id = (uint16_t)(1 + a);
                        v.flags = 0;
                        v.length = 0;
                        if (agg_is_sum1[a]) {
                            v.type = YT_VT_INT64;
                            v.data.bits = g.cnt_nonnull & 0xFFFFFFFFULL;
                        } else if ((g.cnt_nonnull >> 32) == 0) {
                            v.type = YT_VT_NULL;
                            v.data.bits = 0;
                        } else {
                            v.type = val_is_double ? YT_VT_DOUBLE : YT_VT_INT64;
                            v.data.bits = g.sum_bits;
                        }
                    }
                }
            };
            {
                int nt = (int)std::thread::hardware_concurrency();
                if (nt < 1) nt = 1;
                if (nt > 64) nt = 64;
                if (ng < (1 << 16)) nt = 1;
                std::vector<std::thread> ths;
                for (int s2 = 0; s2 < kSlices; s2++) {
                    int64_t sb = (int64_t)s2 * per_sl;
                    int64_t se = std::min<int64_t>(ng, sb + per_sl);
                    if (sb >= se) break;
                    if (sl_ev[s2]) HIP_CHECK(hipEventSynchronize(sl_ev[s2]));
                    if (nt == 1) {
                        emit_range(sb, se);
                        continue;
                    }
                    int kt = nt / kSlices > 0 ? nt / kSlices : 1;
                    int64_t per = (se - sb + kt - 1) / kt;
                    for (int t = 0; t < kt; t++) {
                        int64_t b = sb + (int64_t)t * per, e = b + per;
                        if (b >= se) break;
                        if (e > se) e = se;
                        ths.emplace_back(emit_range, b, e);
                    }
                }
                for (auto& t : ths) t.join();
                if (pool_thr.joinable()) pool_thr.join();
                for (auto& ev : sl_ev) if (ev) (void)hipEventDestroy(ev);
                g_pool.put(stage);
            }
            output->row_count = ng;
            if (has_null_row) {
                YtValue* dst = output->values + output->row_count * ncols;
                dst[0].id = 0;
                dst[0].flags = 0;
                dst[0].type = YT_VT_NULL;
                dst[0].length = 0;
                dst[0].data.bits = 0;
                uint64_t cnt = th.side_cnt[1];
                uint64_t sum_bits = sum_slot >= 0 ? th.side_agg[1][2 * sum_slot] : 0;
                uint64_t nonnull = sum_slot >= 0 ? th.side_agg[1][2 * sum_slot + 1] : 0;
                for (int a = 0; a < plan->agg_count; a++) {
                    YtValue& v = dst[1 + a];
                    v.id = (uint16_t)(1 + a);
                    v.flags = 0;
                    v.length = 0;
                    if (plan->aggs[a]->func == YT_AGG_SUM1) {
                        v.type = YT_VT_INT64;
                        v.data.bits = cnt;
                    } else if (nonnull == 0) {
                        v.type = YT_VT_NULL;
                        v.data.bits = 0;
                    } else {
                        v.type = val_is_double ? YT_VT_DOUBLE : YT_VT_INT64;
                        v.data.bits = sum_bits;
                    }
                }
                output->row_count++;
            }
            goto emitted;
        }
        {
        std::vector<char> pool(hctr[1] ? hctr[1] : 1);
        if (hctr[1]) {
            HIP_CHECK(hipMemcpy(pool.data(), d_pool, hctr[1], hipMemcpyDeviceToHost));
        }
        if (ngroups > 0) {
            /* the fast path streams hgroups in slices; this limited path
             * copies them in one go */
            HIP_CHECK(hipMemcpy(hgroups, d_out, sizeof(OutStrGroup) * ngroups,
                                hipMemcpyDeviceToHost));
        }
        for (int64_t gI = 0; gI < ngroups + 2; gI++) {
            const char* kstr = nullptr;
            uint32_t klen = 0;
            int knull = 0;
            uint64_t cnt, sum_bits, nonnull;
            if (gI < ngroups) {
                const OutStrGroup& g = hgroups[gI];
                kstr = pool.data() + (g.off_len >> 24);
                klen = (uint32_t)(g.off_len & 0xFFFFFF);
                cnt = g.cnt_nonnull & 0xFFFFFFFFULL;
                sum_bits = g.sum_bits;
                nonnull = g.cnt_nonnull >> 32;
            } else {
                int side = (int)(gI - ngroups);
                if (side == 0) continue;                /* no sentinel for strings */
                if (!th.side_used[1]) continue;
                knull = 1;
                cnt = th.side_cnt[1];
                sum_bits = sum_slot >= 0 ? th.side_agg[1][2 * sum_slot] : 0;
                nonnull = sum_slot >= 0 ? th.side_agg[1][2 * sum_slot + 1] : 0;
            }
            if (options->output_row_limit > 0 &&
                output->row_count >= options->output_row_limit) {
                out_limited = 1;
                break;
            }
            if (output->row_count >= output->capacity_rows) {
                rc = YT_ERR_CAPACITY;
                break;
            }
            YtValue* dst = output->values + output->row_count * ncols;
            if (knull) {
                dst[0].type = YT_VT_NULL;
                dst[0].length = 0;
                dst[0].data.bits = 0;
            } else {
                if (output->string_pool_used + klen > output->string_pool_capacity) {
                    rc = YT_ERR_CAPACITY;
                    set_err(errbuf, errlen, "string pool too small");
                    break;
                }
                memcpy(output->string_pool + output->string_pool_used, kstr, klen);
                dst[0].type = YT_VT_STRING;
                dst[0].length = klen;
                dst[0].data.str = output->string_pool + output->string_pool_used;
                output->string_pool_used += klen;
            }
            dst[0].id = 0;
            dst[0].flags = 0;
            for (int a = 0; a < plan->agg_count; a++) {
                YtValue& v = dst[1 + a];
                v.id = (uint16_t)(1 + a);
                v.flags = 0;
                v.length = 0;
                if (plan->aggs[a]->func == YT_AGG_SUM1) {
                    v.type = YT_VT_INT64;
                    v.data.bits = cnt;
                } else if (nonnull == 0) {
                    v.type = YT_VT_NULL;
                    v.data.bits = 0;
                } else {
                    v.type = val_is_double ? YT_VT_DOUBLE : YT_VT_INT64;
                    v.data.bits = sum_bits;
                }
            }
            output->row_count++;
        }
        }
emitted:
        if (stats) {
            stats->rows_read = chunk->row_count;
            stats->rows_written = output->row_count;
            stats->grouped_row_count = ngroups + (th.side_used[1] ? 1 : 0);
            stats->incomplete_input = in_clamped;
            stats->incomplete_output = out_limited || (th.overflow == 2);
            stats->kernel_scan_ms += ms;
            stats->kernel_scan_launches += 1;
            stats->kernel_other_ms += ms_other;   /* compact + result D2H */
            stats->execute_time_ms = now_ms() - tw0;
        }
        g_pool.put(d_accbase); g_pool.put(d_acc); g_pool.put(d_hashes);
        g_pool.put(d_idents); g_pool.put(d_pfxs);
        g_pool.put(d_slots); g_pool.put(d_out); g_pool.put(d_pool);
        g_pool.put(d_ctr); g_pool.put(hgroups);
    }
    return rc;
fail:
    return rc;
}

/* ------------------------------------------------------------------ */
/* public entries                                                      */

extern "C" int yt_gpu_versioned_read(
    const YtVersionedColumn* col, uint64_t timestamp,
    uint64_t* out_bits, uint8_t* out_null, uint8_t* out_visible,
    uint8_t* out_agg, uint64_t stream, char* errbuf, size_t errlen)
{
    int rc = yt_gpu_available(errbuf, errlen);
    if (rc != YT_OK) return rc;
    hipStream_t st = (hipStream_t)(uintptr_t)stream;
    int nseg = col->ts_seg_count;
    if (nseg == 0) return YT_OK;
    if (col->val_seg_count != nseg) {
        set_err(errbuf, errlen, "versioned: segment count mismatch");
        return YT_ERR_INVALID_CHUNK;
    }
    std::vector<VSegDev> h_segs((size_t)nseg);
    std::vector<void*> blobs;
    int64_t total_rows = 0;
    {
        for (int i = 0; i < nseg; i++) {
            const YtTimestampSeg& T = col->ts_segs[i];
            const YtVersionedValueSeg& V = col->val_segs[i];
            char* d_t = nullptr;
            char* d_v = nullptr;
            /* +8 pad: bp_gl may read one word past the last vector */
            HIP_CHECK(pool_alloc(&d_t, (size_t)T.data_size + 8));
            HIP_CHECK(pool_alloc(&d_v, (size_t)V.data_size + 8));
            blobs.push_back(d_t);
            blobs.push_back(d_v);
            HIP_CHECK(hipMemcpyAsync(d_t, T.data, (size_t)T.data_size,
                                     hipMemcpyHostToDevice, st));
            HIP_CHECK(hipMemcpyAsync(d_v, V.data, (size_t)V.data_size,
                                     hipMemcpyHostToDevice, st));
            VSegDev& S = h_segs[i];
            S.row_start = total_rows;
            S.row_count = T.row_count;
            S.base_timestamp = T.base_timestamp;
            S.exp_w = T.expected_writes_per_row;
            S.exp_d = T.expected_deletes_per_row;
            S.exp_v = V.expected_values_per_row;
            S.vtype = V.type;
            S.vflags = V.flags;
            S.base_value = V.base_value;
            S.ts_data = d_t;
            S.val_data = d_v;
            total_rows += T.row_count;
        }
        VSegDev* d_segs = nullptr;
        HIP_CHECK(pool_alloc(&d_segs, sizeof(VSegDev) * nseg));
        blobs.push_back(d_segs);
        HIP_CHECK(hipMemcpyAsync(d_segs, h_segs.data(), sizeof(VSegDev) * nseg,
                                 hipMemcpyHostToDevice, st));
        HIP_CHECK(ytql_launch_versioned_read(d_segs, nseg, total_rows,
                                             timestamp, out_bits, out_null,
                                             out_visible, out_agg, st));
        HIP_CHECK(hipStreamSynchronize(st));
    }
    for (void* p : blobs) g_pool.put(p);
    return YT_OK;
fail:
    for (void* p : blobs) g_pool.put(p);
    return rc;
}

namespace {
struct ScanChunkHandle {
    char* blob = nullptr;              /* device */
    YtSegment* segs = nullptr;         /* host heap */
    YtColumn* col = nullptr;           /* host heap */
};
} /* namespace */

extern "C" int yt_gpu_versioned_scan_table(
    const YtVersionedColumn* const* vcols, int nvcols,
    const YtChunk* key_chunk, uint64_t timestamp,
    YtChunk* out_chunk, void** out_handle,
    uint64_t stream, char* errbuf, size_t errlen);

extern "C" int yt_gpu_versioned_scan_chunk(
    const YtVersionedColumn* col, uint64_t timestamp,
    YtChunk* out_chunk, void** out_handle,
    uint64_t stream, char* errbuf, size_t errlen)
{
    /* single-column case of the table bridge below */
    const YtVersionedColumn* arr[1] = { col };
    return yt_gpu_versioned_scan_table(arr, 1, nullptr, timestamp,
                                       out_chunk, out_handle, stream,
                                       errbuf, errlen);
}

/* Versioned TABLE bridge: N versioned value
 * columns sharing one timestamp layout + optional unversioned KEY columns
 * (keys in the scan format are plain unversioned segments —
 * rowset_builder.cpp key readers), read at T and compacted into one
 * device-resident unversioned chunk [keys..., values...] that
 * yt_gpu_query_execute scans directly. Int64 and double value columns. */
extern "C" int yt_gpu_versioned_scan_table(
    const YtVersionedColumn* const* vcols, int nvcols,
    const YtChunk* key_chunk, uint64_t timestamp,
    YtChunk* out_chunk, void** out_handle,
    uint64_t stream, char* errbuf, size_t errlen)
{
    int rc = yt_gpu_available(errbuf, errlen);
    if (rc != YT_OK) return rc;
    hipStream_t st = (hipStream_t)(uintptr_t)stream;
    memset(out_chunk, 0, sizeof(*out_chunk));
    *out_handle = nullptr;
    if (nvcols < 1 || !vcols) { set_err(errbuf, errlen, "scan_table: need >=1 versioned column"); return YT_ERR_INVALID_PLAN; }

    int64_t n = 0;
    for (int i = 0; i < vcols[0]->ts_seg_count; i++)
        n += vcols[0]->ts_segs[i].row_count;
    for (int j = 1; j < nvcols; j++) {
        int64_t nj = 0;
        for (int i = 0; i < vcols[j]->ts_seg_count; i++)
            nj += vcols[j]->ts_segs[i].row_count;
        if (nj != n) { set_err(errbuf, errlen, "scan_table: column row counts differ"); return YT_ERR_INVALID_CHUNK; }
    }
    const int nkey = key_chunk ? key_chunk->column_count : 0;
    if (key_chunk && key_chunk->row_count != n) {
        set_err(errbuf, errlen, "scan_table: key chunk row count differs");
        return YT_ERR_INVALID_CHUNK;
    }
    for (int c = 0; c < nkey; c++) {
        if (key_chunk->columns[c].value_type != YT_VT_INT64) {
            set_err(errbuf, errlen, "scan_table: int64 key columns this round");
            return YT_ERR_UNSUPPORTED;
        }
    }
    for (int j = 0; j < nvcols; j++) {
        for (int i = 0; i < vcols[j]->val_seg_count; i++) {
            if (vcols[j]->val_segs[i].type >= YT_VSEG_STR_DIRECT_DENSE) {
                set_err(errbuf, errlen,
                        "scan_table: versioned string value columns are "
                        "readable (yt_gpu_versioned_read) but not yet "
                        "bridged into engine chunks");
                return YT_ERR_UNSUPPORTED;
            }
        }
    }
    const int ncols = nkey + nvcols;

    auto* H = new ScanChunkHandle();
    std::vector<uint64_t*> d_bits(ncols, nullptr);
    std::vector<uint8_t*> d_null(ncols, nullptr);
    uint8_t* d_vis = nullptr;
    uint8_t* d_vis2 = nullptr;
    unsigned long long* d_blk = nullptr;
    int64_t* d_off = nullptr;
    const int64_t kSegCap = 128 * 1024;
    int grid = (int)((n + 255) / 256);
    if (grid > 2048) grid = 2048;
    if (grid < 1) grid = 1;
    auto cleanup = [&]() {
        for (auto* p : d_bits) g_pool.put(p);
        for (auto* p : d_null) g_pool.put(p);
        g_pool.put(d_vis); g_pool.put(d_vis2);
        g_pool.put(d_blk); g_pool.put(d_off);
    };
    std::vector<int> col_is_dbl(ncols, 0);
    std::vector<unsigned long long> blk((size_t)grid);
    std::vector<int64_t> seg_rows, seg_bytes;
    int64_t total = 0, col_bytes = 0, blob_bytes = 0;
    int nseg = 0;
    if (n == 0) { *out_handle = H; return YT_OK; }

    for (int c = 0; c < ncols; c++) {
        HIP_CHECK(pool_alloc(&d_bits[c], sizeof(uint64_t) * n));
        HIP_CHECK(pool_alloc(&d_null[c], (size_t)n));
    }
    HIP_CHECK(pool_alloc(&d_vis, (size_t)n));
    HIP_CHECK(pool_alloc(&d_vis2, (size_t)n));
    HIP_CHECK(pool_alloc(&d_blk, sizeof(unsigned long long) * grid));

    /* key columns: decode unversioned DirectDense int64 */
    if (nkey) {
        DeviceRun R2;
        R2.stream = st;
        unsigned maxw = 0;
        int clamped = 0;
        rc = setup_chunk(key_chunk, &R2, &maxw, 0, &clamped, errbuf, errlen);
        if (rc != YT_OK) { cleanup(); delete H; return rc; }
        HIP_CHECK(hipMemsetAsync(R2.d_err, 0, sizeof(unsigned), st));
        for (int c = 0; c < nkey; c++) {
            HIP_CHECK(ytql_launch_unvcol_to_arrays(R2.d_segs, R2.d_segex,
                                                   R2.h_off[c], R2.h_cnt[c], n,
                                                   d_bits[c], d_null[c],
                                                   R2.d_err, st));
        }
        unsigned kerr = 0;
        HIP_CHECK(hipMemcpy(&kerr, R2.d_err, sizeof(unsigned), hipMemcpyDeviceToHost));
        if (kerr) {
            set_err(errbuf, errlen, "scan_table: string key columns not bridged this round");
            cleanup(); delete H;
            return YT_ERR_UNSUPPORTED;
        }
    }
    /* value columns: read at T (col 0 supplies the visibility mask — the
     * write/delete timestamp lists are row-level, shared by every column) */
    for (int j = 0; j < nvcols; j++) {
        const YtVersionedColumn* vc = vcols[j];
        int dbl = 0;
        for (int i = 0; i < vc->val_seg_count; i++)
            if (vc->val_segs[i].type >= YT_VSEG_DOUBLE_DENSE) dbl = 1;
        col_is_dbl[nkey + j] = dbl;
        rc = yt_gpu_versioned_read(vc, timestamp, d_bits[nkey + j],
                                   d_null[nkey + j], j == 0 ? d_vis : d_vis2,
                                   nullptr, stream, errbuf, errlen);
        if (rc != YT_OK) { cleanup(); delete H; return rc; }
    }

    HIP_CHECK(ytql_launch_vis_count(d_vis, n, d_blk, grid, st));
    HIP_CHECK(hipMemcpy(blk.data(), d_blk, sizeof(unsigned long long) * grid,
                        hipMemcpyDeviceToHost));
    {
        unsigned long long run = 0;
        for (int i = 0; i < grid; i++) { unsigned long long c = blk[i]; blk[i] = run; run += c; }
        total = (int64_t)run;
    }
    nseg = (int)((total + kSegCap - 1) / kSegCap);
    if (total == 0) {
        cleanup();
        *out_handle = H;
        return YT_OK;
    }
    HIP_CHECK(hipMemcpyAsync(d_blk, blk.data(), sizeof(unsigned long long) * grid,
                             hipMemcpyHostToDevice, st));

    /* one blob: per column, per segment [header][values w64][null bitmap] */
    seg_rows.resize(nseg);
    seg_bytes.resize(nseg);
    for (int i = 0; i < nseg; i++) {
        seg_rows[i] = i + 1 < nseg ? kSegCap : total - (int64_t)i * kSegCap;
        int64_t bm = (((seg_rows[i] + 7) / 8) + 7) & ~(int64_t)7;
        seg_bytes[i] = 8 + seg_rows[i] * 8 + bm;
        col_bytes += seg_bytes[i];
    }
    blob_bytes = col_bytes * ncols;
    HIP_CHECK(pool_alloc(&H->blob, (size_t)blob_bytes + 8));
    HIP_CHECK(hipMemsetAsync(H->blob, 0, (size_t)blob_bytes, st));
    HIP_CHECK(pool_alloc(&d_off, sizeof(int64_t) * nseg));

    H->segs = (YtSegment*)calloc((size_t)nseg * ncols, sizeof(YtSegment));
    H->col = (YtColumn*)calloc(ncols, sizeof(YtColumn));
    for (int c = 0; c < ncols; c++) {
        std::vector<int64_t> off(nseg);
        int64_t at = (int64_t)c * col_bytes;
        for (int i = 0; i < nseg; i++) {
            off[i] = at;
            /* int: bitpack header rows|64<<56; double: [u64 count] */
            uint64_t hdr = col_is_dbl[c] ? (uint64_t)seg_rows[i]
                                         : ((uint64_t)seg_rows[i] | (64ULL << 56));
            HIP_CHECK(hipMemcpyAsync(H->blob + at, &hdr, 8,
                                     hipMemcpyHostToDevice, st));
            YtSegment& sg = H->segs[c * nseg + i];
            sg.type = col_is_dbl[c] ? YT_SEG_DOUBLE : YT_SEG_DIRECT_DENSE;
            sg.row_count = (int32_t)seg_rows[i];
            sg.min_value = 0;
            sg.data = H->blob + at;
            sg.data_size = seg_bytes[i];
            at += seg_bytes[i];
        }
        HIP_CHECK(hipMemcpyAsync(d_off, off.data(), sizeof(int64_t) * nseg,
                                 hipMemcpyHostToDevice, st));
        HIP_CHECK(ytql_launch_vis_scatter(d_vis, d_null[c], d_bits[c], n, d_blk,
                                          (uint64_t)kSegCap, d_off,
                                          H->blob, col_is_dbl[c], grid, st));
        HIP_CHECK(hipStreamSynchronize(st));
        YtColumn& C2 = H->col[c];
        C2.value_type = col_is_dbl[c] ? YT_VT_DOUBLE : YT_VT_INT64;
        C2.segment_count = nseg;
        C2.segments = &H->segs[c * nseg];
    }
    out_chunk->row_count = total;
    out_chunk->column_count = ncols;
    out_chunk->columns = H->col;
    *out_handle = H;
    cleanup();
    return YT_OK;
fail:
    cleanup();
    delete H;
    return rc;
}

extern "C" void yt_gpu_scan_chunk_free(YtChunk* chunk, void* handle)
{
    auto* H = (ScanChunkHandle*)handle;
    if (!H) return;
    g_pool.put(H->blob);
    free(H->segs);
    free(H->col);
    delete H;
    if (chunk) memset(chunk, 0, sizeof(*chunk));
}

extern "C" int yt_gpu_query_execute(
    const YtPlan* plan, const YtChunk* chunk, const YtExecOptions* options,
    YtRowset* output, YtStatistics* stats, char* errbuf, size_t errlen)
{
    int rc = yt_gpu_available(errbuf, errlen);
    if (rc != YT_OK) return rc;
    if (!plan || !chunk || !output) { set_err(errbuf, errlen, "null argument"); return YT_ERR_INVALID_PLAN; }

    double tw0 = now_ms();
    double tp0 = 0, tp1 = 0, tp2 = 0;   /* phase checkpoints (YTQL_TIMING) */
    YtExecOptions defopt;
    memset(&defopt, 0, sizeof(defopt));
    if (!options) options = &defopt;
    if (stats) memset(stats, 0, sizeof(*stats));

    if (plan->agg_count == 0 && plan->project_count == 0 &&
        plan->key_count == 0) {
        set_err(errbuf, errlen, "empty plan");
        return YT_ERR_INVALID_PLAN;
    }
    output->totals_row = 0;
    if (plan->with_totals || plan->having) {
        if (plan->key_count == 0) {
            set_err(errbuf, errlen, "WITH TOTALS / HAVING requires GROUP BY");
            return YT_ERR_INVALID_PLAN;
        }
        if (plan->project_count) {
            set_err(errbuf, errlen,
                    "WITH TOTALS / HAVING with projections: not this round");
            return YT_ERR_UNSUPPORTED;
        }
        if (options->output_row_limit > 0) {
            set_err(errbuf, errlen,
                    "WITH TOTALS / HAVING with OutputRowLimit: not this round");
            return YT_ERR_UNSUPPORTED;
        }
    }

    {
        int kc;
        if (plan->key_count == 1 && plan->agg_count > 0 &&
            expr_is_col(plan->keys[0], &kc) && kc < chunk->column_count &&
            chunk->columns[kc].value_type == YT_VT_STRING) {
            if (plan->join) {
                set_err(errbuf, errlen,
                        "join with string group keys: not this round");
                return YT_ERR_UNSUPPORTED;
            }
            rc = run_string_group(plan, chunk, options, kc, output, stats,
                                  tw0, errbuf, errlen);
            if (rc == YT_OK && (plan->order_count > 0 || plan->with_totals ||
                                plan->having)) {
                rc = finish_output(plan, output, errbuf, errlen);
                if (rc == YT_OK && stats)
                    stats->rows_written = output->row_count;
            }
            return rc;
        }
    }

    DevPlan dp;
    rc = build_devplan(plan, chunk, &dp, errbuf, errlen);
    if (rc) return rc;

    JoinRun JR[2];
    const JoinDev* jd = nullptr;
    const JoinDev* jd2 = nullptr;
    rc = setup_join_chain(plan, chunk, JR, &jd, &jd2,
                          (hipStream_t)(uintptr_t)options->stream,
                          errbuf, errlen);
    if (rc) return rc;
    if (jd->active && jd->has_dups &&
        (plan->order_count > 0 || plan->agg_count == 0)) {
        /* the ORDER BY / scan-project machinery tracks candidates by
         * primary row id; a one-to-many join breaks that identity —
         * grouped plans take the cross-product path instead */
        set_err(errbuf, errlen,
                "join: duplicate foreign keys with ORDER BY / plain scan "
                "not this round (GROUP BY plans supported)");
        return YT_ERR_UNSUPPORTED;
    }

    if (plan->agg_count == 0 && plan->key_count == 0) {
        if (plan->order_count > 0) {
            /* scan + ORDER BY ... LIMIT: k-selection, no full materialization */
            return run_scan_topk(plan, chunk, options, &dp, jd, jd2, output,
                                 stats, tw0, errbuf, errlen);
        }
        return run_scan_project(plan, chunk, options, &dp, jd, jd2, output,
                                stats, tw0, errbuf, errlen);
    }

    FastShape fs;
    analyze_fast(plan, chunk, &fs);

    DeviceRun R;
    R.stream = (hipStream_t)(uintptr_t)options->stream;
    if (options->device) {
        hipError_t e = hipSetDevice(options->device);
        if (e != hipSuccess) { set_err(errbuf, errlen, "bad device"); return YT_ERR_HIP; }
    }

    unsigned maxw = 0;
    int in_clamped = 0;
    rc = setup_chunk(chunk, &R, &maxw, options->input_row_limit, &in_clamped,
                     errbuf, errlen);
    if (rc) return rc;
    if (dp.kp_count && chunk->row_count > 0 && R.nsegs > 0) {
        rc = pack_group_key(&dp, &R, errbuf, errlen);
        if (rc) return rc;
    }

    if (chunk->row_count == 0 || R.nsegs == 0) {
        /* zero groups -> single final flush emits nothing (registry.cpp:1481) */
        output->row_count = 0;
        output->column_count = plan->project_count ? plan->project_count
            : (plan->key_count + plan->agg_count);
        return YT_OK;
    }

    rc = setup_table(&R, plan->agg_count,
                     options->max_groups_hint,
                     options->group_row_limit, errbuf, errlen);
    if (rc) return rc;

    const bool timing = getenv("YTQL_TIMING") != nullptr;   /* phase breakdown to stderr */
    tp0 = now_ms();
    rc = run_scan(plan, chunk, options, &R, &dp, jd, jd2, &fs, maxw, stats,
                  errbuf, errlen);
    if (rc) return rc;
    tp1 = now_ms();

    /* compact + readback (pinned staging from the pool) */
    TableHdr th;
    OutGroup* hgroups = nullptr;
    int64_t ngroups = 0;
    {
        HIP_CHECK(hipMemcpy(&th, R.d_th, sizeof(th), hipMemcpyDeviceToHost));
        if (th.overflow == 1) { set_err(errbuf, errlen, "group table overflow — raise max_groups_hint"); rc = YT_ERR_CAPACITY; goto fail; }
        ngroups = (int64_t)th.ngroups;
        if (ngroups > 0 && !R.groups_compacted) {
            HIP_CHECK(pool_alloc(&R.d_groups, sizeof(OutGroup) * ngroups));
            HIP_CHECK(pool_alloc(&R.d_counter, sizeof(unsigned long long)));
            HIP_CHECK(hipMemsetAsync(R.d_counter, 0, sizeof(unsigned long long), R.stream));
            HIP_CHECK(ytql_launch_compact(nullptr, R.d_th, R.d_slots, plan->agg_count,
                                          R.d_groups, R.d_counter, R.nslots, R.stream));
        }
        if (ngroups > 0) {
            HIP_CHECK(pool_alloc_host(&hgroups, sizeof(OutGroup) * ngroups));
            HIP_CHECK(hipMemcpyAsync(hgroups, R.d_groups, sizeof(OutGroup) * ngroups,
                                     hipMemcpyDeviceToHost, R.stream));
            HIP_CHECK(hipStreamSynchronize(R.stream));
        }
    }
    tp2 = now_ms();
    {
        std::vector<uint64_t> gaccum(1 + 2 * kMaxAggs, 0);
        if (fs.valid && fs.key_col < 0) {
            HIP_CHECK(hipMemcpy(gaccum.data(), R.d_gaccum,
                                sizeof(uint64_t) * (1 + 2 * kMaxAggs), hipMemcpyDeviceToHost));
        }
        int out_limited = 0;
        rc = emit_rows(plan, chunk, &dp, hgroups, ngroups, th, 0, gaccum.data(),
                       fs.valid && fs.key_col < 0,
                       options->output_row_limit, &out_limited,
                       output, errbuf, errlen);
        g_pool.put(hgroups);
        if (rc) return rc;
        if (out_limited && stats) stats->incomplete_output = 1;
    }
    if (timing) {
        double tp3 = now_ms();
        fprintf(stderr, "[ytql timing] pre-scan %.2fms scan-wall %.2fms compact+D2H %.2fms emit %.2fms\n",
                tp0 - tw0, tp1 - tp0, tp2 - tp1, tp3 - tp2);
    }
    if (plan->order_count > 0 || plan->with_totals || plan->having) {
        /* grouped output is already bounded: order/totals/having on the host
         * (combined group+order mode, registry.cpp:1677-1699) */
        rc = finish_output(plan, output, errbuf, errlen);
        if (rc) return rc;
        if (stats) stats->rows_written = output->row_count;
    }
    if (stats) {
        stats->rows_read = (options->input_row_limit > 0 &&
                            options->input_row_limit < chunk->row_count)
            ? options->input_row_limit : chunk->row_count;
        stats->incomplete_input = in_clamped;
        int64_t bytes = 0;
        for (int c = 0; c < chunk->column_count; c++)
            for (int s = 0; s < chunk->columns[c].segment_count; s++)
                bytes += chunk->columns[c].segments[s].data_size;
        stats->data_weight_read = bytes;
        stats->rows_written = output->row_count;
        stats->grouped_row_count = (int64_t)th.ngroups
            + (plan->key_count ? (int64_t)(th.side_used[0] + th.side_used[1]) : 0);
        stats->incomplete_output |= (th.overflow == 2);
        stats->execute_time_ms = now_ms() - tw0;
    }
    return YT_OK;
fail:
    return rc;
}

static int query_partial_impl(
    const YtPlan* plan, const YtChunk* chunk, const YtExecOptions* options,
    int32_t partition_count, void* states_device, int64_t capacity_rows,
    int64_t* part_counts,
    const uint64_t* key_zzmin, const uint64_t* key_zzmax,
    YtStatistics* stats, char* errbuf, size_t errlen)
{
    int rc = yt_gpu_available(errbuf, errlen);
    if (rc != YT_OK) return rc;
    if (plan->key_count < 1 || plan->key_count > kMaxPackKeys) {
        set_err(errbuf, errlen, "partial: 1..4 group keys");
        return YT_ERR_UNSUPPORTED;
    }
    if (plan->key_count > 1 && (!key_zzmin || !key_zzmax)) {
        set_err(errbuf, errlen,
                "partial: multi-key sharding needs the common key ranges "
                "(yt_gpu_key_ranges + cross-rank min/max reduce)");
        return YT_ERR_UNSUPPORTED;
    }
    if (partition_count < 1 || partition_count > 64) { set_err(errbuf, errlen, "partial: 1..64 partitions"); return YT_ERR_UNSUPPORTED; }
    /* join at the bottom query (the reference's coordinated split keeps
     * JoinClause in the bottom, coordinator.cpp:130-170: the dimension
     * table is available on every node) — each rank builds the foreign
     * hash table from its own copy of the dimension chunk and the generic
     * scan probes it while producing partial states. Multi-key packing
     * would need zigzag ranges of foreign columns; deferred. */
    if (plan->join && plan->key_count > 1) {
        set_err(errbuf, errlen,
                "partial: join with multi-key GROUP BY not this round");
        return YT_ERR_UNSUPPORTED;
    }
    /* the 32-byte state carries ONE running {sum, nonnull-count} slot plus
     * the row count: sum(x) or avg(x) (the reference's coordinated avg is
     * exactly {count,sum}: GroupByWithAvgCoordinated ql_query_ut.cpp:2760) */
    int sum_slot = -1;
    int state_func = YT_AGG_SUM;
    for (int a = 0; a < plan->agg_count; a++) {
        int f = plan->aggs[a]->func;
        if (f == YT_AGG_SUM || f == YT_AGG_AVG || f == YT_AGG_MIN ||
            f == YT_AGG_MAX || f == YT_AGG_FIRST) {
            if (sum_slot >= 0) { set_err(errbuf, errlen, "partial: one value-carrying agg max this round"); return YT_ERR_UNSUPPORTED; }
            sum_slot = a;
            state_func = f;
        } else if (f != YT_AGG_SUM1) {
            set_err(errbuf, errlen, "partial: sum/avg/min/max/first/sum(1) only");
            return YT_ERR_UNSUPPORTED;
        }
    }

    YtExecOptions defopt;
    memset(&defopt, 0, sizeof(defopt));
    if (!options) options = &defopt;
    if (stats) memset(stats, 0, sizeof(*stats));

    double tw0 = now_ms();
    DevPlan dp;
    rc = build_devplan(plan, chunk, &dp, errbuf, errlen);
    if (rc) return rc;
    FastShape fs;
    analyze_fast(plan, chunk, &fs);

    DeviceRun R;
    R.stream = (hipStream_t)(uintptr_t)options->stream;

    unsigned maxw = 0;
    int in_clamped = 0;
    rc = setup_chunk(chunk, &R, &maxw, options->input_row_limit, &in_clamped,
                     errbuf, errlen);
    if (rc) return rc;
    if (chunk->row_count == 0 || R.nsegs == 0) {
        for (int p = 0; p < partition_count; p++) part_counts[p] = 0;
        return YT_OK;
    }
    if (dp.kp_count && key_zzmin && key_zzmax) {
        /* pack the composite with the COMMON cross-rank ranges so every
         * rank's key_bits agree (partition hash + merge identity) */
        for (int i = 0; i < dp.kp_count; i++) {
            int c = dp.kp_col[i];
            R.col_zzmin[c] = key_zzmin[i];
            R.col_zzmax[c] = key_zzmax[i];
        }
        rc = pack_group_key(&dp, &R, errbuf, errlen);
        if (rc) return rc;
    } else if (dp.kp_count) {
        rc = pack_group_key(&dp, &R, errbuf, errlen);
        if (rc) return rc;
    }
    rc = setup_table(&R, plan->agg_count, options->max_groups_hint,
                     options->group_row_limit, errbuf, errlen);
    if (rc) return rc;
    double tq0 = now_ms();
    JoinRun JRp[2];
    const JoinDev* pjd = nullptr;
    const JoinDev* pjd2 = nullptr;
    rc = setup_join_chain(plan, chunk, JRp, &pjd, &pjd2,
                          (hipStream_t)(uintptr_t)options->stream,
                          errbuf, errlen);
    if (rc) return rc;
    /* dup foreign keys on item 0 are fine here: a partial is always grouped */
    rc = run_scan(plan, chunk, options, &R, &dp, pjd, pjd2, &fs, maxw, stats,
                  errbuf, errlen);
    if (rc) return rc;
    if (getenv("YTQL_TIMING"))
        fprintf(stderr, "[ytql timing] partial: setup %.2fms scan %.2fms\n",
                tq0 - tw0, now_ms() - tq0);

    TableHdr th;
    double tq15 = now_ms();
    {
        HIP_CHECK(hipMemcpy(&th, R.d_th, sizeof(th), hipMemcpyDeviceToHost));
        if (th.overflow == 1) { set_err(errbuf, errlen, "group table overflow"); rc = YT_ERR_CAPACITY; goto fail; }
        int64_t ngroups = (int64_t)th.ngroups;
        int64_t total = ngroups + th.side_used[0] + th.side_used[1];
        if (total > capacity_rows) { set_err(errbuf, errlen, "state buffer too small"); rc = YT_ERR_CAPACITY; goto fail; }

        if (!R.groups_compacted) {
            /* compact table into OutGroups (device) */
            HIP_CHECK(pool_alloc(&R.d_groups, sizeof(OutGroup) * (total ? total : 1)));
            HIP_CHECK(pool_alloc(&R.d_counter, sizeof(unsigned long long)));
            HIP_CHECK(hipMemsetAsync(R.d_counter, 0, sizeof(unsigned long long), R.stream));
            if (ngroups > 0) {
                HIP_CHECK(ytql_launch_compact(nullptr, R.d_th, R.d_slots, plan->agg_count,
                                              R.d_groups, R.d_counter, R.nslots, R.stream));
            }
        } else if (R.groups_capacity < total) {
            set_err(errbuf, errlen, "state buffer too small");
            rc = YT_ERR_CAPACITY;
            goto fail;
        }
        /* append side groups from host */
        int64_t pos = ngroups;
        for (int side = 0; side < 2; side++) {
            if (!th.side_used[side]) continue;
            OutGroup g;
            memset(&g, 0, sizeof(g));
            g.key_bits = th.side_key_bits[side];
            g.key_meta = (side == 1);
            g.cnt = th.side_cnt[side];
            for (int a = 0; a < plan->agg_count; a++) {
                g.agg_bits[a] = th.side_agg[side][2 * a];
                g.agg_nonnull[a] = th.side_agg[side][2 * a + 1];
            }
            HIP_CHECK(hipMemcpyAsync(R.d_groups + pos, &g, sizeof(g),
                                     hipMemcpyHostToDevice, R.stream));
            pos++;
        }

        /* partition counts → host prefix → scatter */
        double tq2 = now_ms();
        unsigned long long* d_counts = nullptr;
        HIP_CHECK(pool_alloc(&d_counts, sizeof(unsigned long long) * partition_count));
        HIP_CHECK(hipMemsetAsync(d_counts, 0, sizeof(unsigned long long) * partition_count, R.stream));
        HIP_CHECK(ytql_launch_part_count(R.d_groups, total, partition_count, sum_slot,
                                         d_counts, R.stream));
        std::vector<unsigned long long> counts(partition_count);
        HIP_CHECK(hipMemcpyAsync(counts.data(), d_counts,
                                 sizeof(unsigned long long) * partition_count,
                                 hipMemcpyDeviceToHost, R.stream));
        HIP_CHECK(hipStreamSynchronize(R.stream));
        std::vector<unsigned long long> cursors(partition_count);
        unsigned long long run = 0;
        for (int p = 0; p < partition_count; p++) { cursors[p] = run; run += counts[p]; }
        HIP_CHECK(hipMemcpyAsync(d_counts, cursors.data(),
                                 sizeof(unsigned long long) * partition_count,
                                 hipMemcpyHostToDevice, R.stream));
        int sum_is_double = 0;
        if (sum_slot >= 0) {
            /* dp.col_types already appends the joined foreign columns */
            int at = expr_static_type(plan->aggs[sum_slot]->arg, dp.col_types);
            sum_is_double = at == YT_VT_DOUBLE;
            if ((state_func == YT_AGG_MIN || state_func == YT_AGG_MAX) &&
                at != YT_VT_INT64 && at != YT_VT_DOUBLE) {
                set_err(errbuf, errlen,
                        "partial: min/max over int64/double args this round");
                rc = YT_ERR_UNSUPPORTED;
                goto fail;
            }
        }
        HIP_CHECK(ytql_launch_part_scatter(R.d_groups, total, partition_count, sum_slot,
                                           sum_is_double, state_func,
                                           d_counts, (YtStateRow*)states_device, R.stream));
        HIP_CHECK(hipStreamSynchronize(R.stream));
        g_pool.put(d_counts);
        for (int p = 0; p < partition_count; p++) part_counts[p] = (int64_t)counts[p];
        if (getenv("YTQL_TIMING"))
            fprintf(stderr, "[ytql timing] partial total %.2fms (compact+sides %.2fms count/scatter %.2fms)\n",
                    now_ms() - tw0, tq2 - tq15, now_ms() - tq2);
        if (stats) {
            stats->rows_read = chunk->row_count;
            stats->grouped_row_count = total;
            stats->incomplete_output = (th.overflow == 2);
        }
    }
    return YT_OK;
fail:
    return rc;
}

extern "C" int yt_gpu_query_partial(
    const YtPlan* plan, const YtChunk* chunk, const YtExecOptions* options,
    int32_t partition_count, void* states_device, int64_t capacity_rows,
    int64_t* part_counts, YtStatistics* stats, char* errbuf, size_t errlen)
{
    if (plan && plan->key_count != 1) {
        set_err(errbuf, errlen, "partial: need exactly 1 key (use _mk)");
        return YT_ERR_UNSUPPORTED;
    }
    return query_partial_impl(plan, chunk, options, partition_count,
                              states_device, capacity_rows, part_counts,
                              nullptr, nullptr, stats, errbuf, errlen);
}

extern "C" int yt_gpu_query_partial_mk(
    const YtPlan* plan, const YtChunk* chunk, const YtExecOptions* options,
    int32_t partition_count, void* states_device, int64_t capacity_rows,
    int64_t* part_counts,
    const uint64_t* key_zzmin, const uint64_t* key_zzmax,
    YtStatistics* stats, char* errbuf, size_t errlen)
{
    return query_partial_impl(plan, chunk, options, partition_count,
                              states_device, capacity_rows, part_counts,
                              key_zzmin, key_zzmax, stats, errbuf, errlen);
}

/* STRING-keyed bottom query: the local string group-by (run_string_group)
 * followed by a hash partition of the groups into YtStateRow records +
 * per-partition key-byte pool slices. The caller all-to-alls BOTH buffers
 * (counts and byte counts per partition are returned); each received
 * state's key_bits is an offset within its own slice. Mirrors the
 * reference's key shuffle (shuffling_reader.cpp:40-42) applied to string
 * group keys. */
extern "C" int yt_gpu_query_partial_str(
    const YtPlan* plan, const YtChunk* chunk, const YtExecOptions* options,
    int32_t partition_count, void* states_device, int64_t capacity_rows,
    void* pool_device, int64_t pool_capacity,
    int64_t* part_counts, int64_t* part_pool_bytes,
    YtStatistics* stats, char* errbuf, size_t errlen)
{
    int rc = yt_gpu_available(errbuf, errlen);
    if (rc != YT_OK) return rc;
    if (!plan || !chunk) { set_err(errbuf, errlen, "null argument"); return YT_ERR_INVALID_PLAN; }
    if (partition_count < 1 || partition_count > 64) { set_err(errbuf, errlen, "partial_str: 1..64 partitions"); return YT_ERR_UNSUPPORTED; }
    int kc = -1;
    if (!(plan->key_count == 1 && plan->agg_count > 0 &&
          expr_is_col(plan->keys[0], &kc) && kc < chunk->column_count &&
          chunk->columns[kc].value_type == YT_VT_STRING)) {
        set_err(errbuf, errlen, "partial_str: needs one string key column + aggregates");
        return YT_ERR_UNSUPPORTED;
    }
    if (plan->join) { set_err(errbuf, errlen, "partial_str: join with string keys not this round"); return YT_ERR_UNSUPPORTED; }
    YtExecOptions defopt;
    memset(&defopt, 0, sizeof(defopt));
    if (!options) options = &defopt;
    if (stats) memset(stats, 0, sizeof(*stats));
    StrPartialOut po;
    po.partition_count = partition_count;
    po.states_device = (YtStateRow*)states_device;
    po.capacity_rows = capacity_rows;
    po.pool_device = (char*)pool_device;
    po.pool_capacity = pool_capacity;
    po.part_counts = part_counts;
    po.part_pool_bytes = part_pool_bytes;
    return run_string_group(plan, chunk, options, kc, nullptr, stats,
                            now_ms(), errbuf, errlen, &po);
}

/* per-rank key-column zigzag ranges for the cross-rank reduce (meta-only
 * bound: conservative is fine, the reduce of bounds is a bound) */
extern "C" int yt_gpu_key_ranges(
    const YtPlan* plan, const YtChunk* chunk,
    uint64_t* key_zzmin, uint64_t* key_zzmax,
    uint64_t stream, char* errbuf, size_t errlen)
{
    int rc = yt_gpu_available(errbuf, errlen);
    if (rc != YT_OK) return rc;
    if (plan->key_count < 1 || plan->key_count > kMaxPackKeys) {
        set_err(errbuf, errlen, "key_ranges: 1..4 group keys");
        return YT_ERR_UNSUPPORTED;
    }
    DeviceRun R;
    R.stream = (hipStream_t)(uintptr_t)stream;
    unsigned maxw = 0;
    int cl = 0;
    rc = setup_chunk(chunk, &R, &maxw, 0, &cl, errbuf, errlen);
    if (rc) return rc;
    for (int i = 0; i < plan->key_count; i++) {
        int c;
        if (!expr_is_col(plan->keys[i], &c) || c >= chunk->column_count) {
            set_err(errbuf, errlen, "key_ranges: plain key columns only");
            return YT_ERR_UNSUPPORTED;
        }
        key_zzmin[i] = R.col_zzmin[c];
        key_zzmax[i] = R.col_zzmax[c];
    }
    return YT_OK;
}

static int merge_states_impl(
    const YtPlan* plan, const void* states_device, int64_t state_row_count,
    const uint8_t* col_types,
    const uint64_t* key_zzmin, const uint64_t* key_zzmax,
    const YtExecOptions* options, YtRowset* output, YtStatistics* stats,
    char* errbuf, size_t errlen)
{
    int rc = yt_gpu_available(errbuf, errlen);
    if (rc != YT_OK) return rc;
    if (plan->key_count < 1 || plan->key_count > kMaxPackKeys) {
        set_err(errbuf, errlen, "merge: 1..4 group keys");
        return YT_ERR_UNSUPPORTED;
    }
    if (plan->key_count > 1 && (!key_zzmin || !key_zzmax)) {
        set_err(errbuf, errlen, "merge: multi-key needs the common key ranges");
        return YT_ERR_UNSUPPORTED;
    }
    output->totals_row = 0;
    int sum_slot = -1;
    int state_func = YT_AGG_SUM;
    for (int a = 0; a < plan->agg_count; a++) {
        int f = plan->aggs[a]->func;
        if (f == YT_AGG_SUM || f == YT_AGG_AVG || f == YT_AGG_MIN ||
            f == YT_AGG_MAX || f == YT_AGG_FIRST) { sum_slot = a; state_func = f; }
    }

    YtExecOptions defopt;
    memset(&defopt, 0, sizeof(defopt));
    if (!options) options = &defopt;
    if (stats) memset(stats, 0, sizeof(*stats));

    double tw0 = now_ms();
    DeviceRun R;
    R.stream = (hipStream_t)(uintptr_t)options->stream;
    rc = setup_table(&R, plan->agg_count, options->max_groups_hint, 0, errbuf, errlen);
    if (rc) return rc;
    rc = ensure_slots(&R, plan->agg_count, errbuf, errlen);
    if (rc) return rc;
    double tq1 = now_ms();

    HIP_CHECK(ytql_launch_merge_states((const YtStateRow*)states_device, state_row_count,
                                       plan->agg_count, sum_slot, state_func,
                                       R.d_th, R.d_slots, R.stream));
    HIP_CHECK(hipStreamSynchronize(R.stream));
    if (getenv("YTQL_TIMING"))
        fprintf(stderr, "[ytql timing] merge: setup %.2fms kernel %.2fms\n",
                tq1 - tw0, now_ms() - tq1);
    {
        TableHdr th;
        HIP_CHECK(hipMemcpy(&th, R.d_th, sizeof(th), hipMemcpyDeviceToHost));
        if (th.overflow == 1) { set_err(errbuf, errlen, "merge table overflow"); rc = YT_ERR_CAPACITY; goto fail; }
        int64_t ngroups = (int64_t)th.ngroups;
        OutGroup* hgroups = nullptr;
        if (ngroups > 0) {
            HIP_CHECK(pool_alloc(&R.d_groups, sizeof(OutGroup) * ngroups));
            HIP_CHECK(pool_alloc(&R.d_counter, sizeof(unsigned long long)));
            HIP_CHECK(hipMemsetAsync(R.d_counter, 0, sizeof(unsigned long long), R.stream));
            HIP_CHECK(ytql_launch_compact(nullptr, R.d_th, R.d_slots, plan->agg_count,
                                          R.d_groups, R.d_counter, R.nslots, R.stream));
            HIP_CHECK(pool_alloc_host(&hgroups, sizeof(OutGroup) * ngroups));
            HIP_CHECK(hipMemcpyAsync(hgroups, R.d_groups, sizeof(OutGroup) * ngroups,
                                     hipMemcpyDeviceToHost, R.stream));
            HIP_CHECK(hipStreamSynchronize(R.stream));
        }
        /* fabricate a single-int64-column chunk view for type inference:
         * the key static type is carried by plan->keys[0] over col types.
         * Merge callers use the same chunk schema as partial; we only need
         * col_types for expr_static_type — rebuild from the key expr columns
         * is not possible here, so default key/sum types to int64. */
        YtColumn col;
        memset(&col, 0, sizeof(col));
        col.value_type = YT_VT_INT64;
        col.segment_count = 0;
        std::vector<YtColumn> cols(kMaxCols, col);
        if (col_types) {
            for (int c2 = 0; c2 < kMaxCols; c2++)
                cols[c2].value_type = col_types[c2];
        }
        YtChunk fake;
        fake.row_count = 0;
        fake.column_count = kMaxCols;
        fake.columns = cols.data();
        /* multi-key: rebuild the composite packing from the COMMON ranges
         * (identical math to every rank's partial — pack_group_key) so
         * emit unpacks the key components */
        DevPlan dpk;
        DevPlan* dpk_p = nullptr;
        if (plan->key_count > 1) {
            memset(&dpk, 0, sizeof(dpk));
            dpk.ncols = kMaxCols;
            for (int c2 = 0; c2 < kMaxCols; c2++)
                dpk.col_types[c2] = col_types ? col_types[c2] : YT_VT_INT64;
            dpk.kp_count = plan->key_count;
            DeviceRun R2;   /* host-side ranges only */
            for (int i = 0; i < plan->key_count; i++) {
                int c2;
                if (!expr_is_col(plan->keys[i], &c2) || c2 >= kMaxCols) {
                    set_err(errbuf, errlen, "merge: plain key columns only");
                    g_pool.put(hgroups);
                    return YT_ERR_UNSUPPORTED;
                }
                dpk.kp_col[i] = c2;
                dpk.kp_signed[i] = dpk.col_types[c2] == YT_VT_INT64;
                R2.col_zzmin[c2] = key_zzmin[i];
                R2.col_zzmax[c2] = key_zzmax[i];
            }
            rc = pack_group_key(&dpk, &R2, errbuf, errlen);
            if (rc) { g_pool.put(hgroups); return rc; }
            dpk_p = &dpk;
        }
        int out_limited = 0;
        rc = emit_rows(plan, &fake, dpk_p, hgroups, ngroups, th, 0,
                       nullptr, 0, options->output_row_limit, &out_limited,
                       output, errbuf, errlen);
        g_pool.put(hgroups);
        if (rc) return rc;
        if (out_limited && stats) stats->incomplete_output = 1;
        if (plan->order_count > 0 || plan->with_totals || plan->having) {
            /* ORDER BY / WITH TOTALS / HAVING apply at the front query */
            rc = finish_output(plan, output, errbuf, errlen);
            if (rc) return rc;
        }
        if (stats) {
            stats->rows_written = output->row_count;
            stats->grouped_row_count = ngroups + th.side_used[0] + th.side_used[1];
        }
    }
    return YT_OK;
fail:
    return rc;
}

extern "C" int yt_gpu_merge_states(
    const YtPlan* plan, const void* states_device, int64_t state_row_count,
    const uint8_t* col_types,
    const YtExecOptions* options, YtRowset* output, YtStatistics* stats,
    char* errbuf, size_t errlen)
{
    if (plan && plan->key_count != 1) {
        set_err(errbuf, errlen, "merge: need exactly 1 key (use _mk)");
        return YT_ERR_UNSUPPORTED;
    }
    return merge_states_impl(plan, states_device, state_row_count, col_types,
                             nullptr, nullptr, options, output, stats,
                             errbuf, errlen);
}

extern "C" int yt_gpu_merge_states_mk(
    const YtPlan* plan, const void* states_device, int64_t state_row_count,
    const uint8_t* col_types,
    const uint64_t* key_zzmin, const uint64_t* key_zzmax,
    const YtExecOptions* options, YtRowset* output, YtStatistics* stats,
    char* errbuf, size_t errlen)
{
    return merge_states_impl(plan, states_device, state_row_count, col_types,
                             key_zzmin, key_zzmax, options, output, stats,
                             errbuf, errlen);
}

/* STRING-keyed front query: merge exchanged state rows (key_bits =
 * slice-local pool reference) whose pool slices were concatenated in
 * segment order on the receiver. seg_counts / seg_pool_bytes give each
 * segment's row and byte extent; states and pool are DEVICE buffers. */
extern "C" int yt_gpu_merge_states_str(
    const YtPlan* plan, const void* states_device, const int64_t* seg_counts,
    int32_t nseg_in, const void* pool_device, const int64_t* seg_pool_bytes,
    const uint8_t* col_types,      /* original chunk column types (sum arg) */
    const YtExecOptions* options, YtRowset* output, YtStatistics* stats,
    char* errbuf, size_t errlen)
{
    int rc = yt_gpu_available(errbuf, errlen);
    if (rc != YT_OK) return rc;
    if (!plan || !output || nseg_in < 1) { set_err(errbuf, errlen, "merge_str: bad arguments"); return YT_ERR_INVALID_PLAN; }
    if (plan->key_count != 1) { set_err(errbuf, errlen, "merge_str: one string key"); return YT_ERR_UNSUPPORTED; }
    if (plan->with_totals || plan->having || plan->order_count) {
        set_err(errbuf, errlen, "merge_str: totals/having/order not this round");
        return YT_ERR_UNSUPPORTED;
    }
    int sum_slot = -1;
    for (int a = 0; a < plan->agg_count; a++) {
        if (plan->aggs[a]->func == YT_AGG_SUM) {
            if (sum_slot >= 0) { set_err(errbuf, errlen, "merge_str: one sum agg max"); return YT_ERR_UNSUPPORTED; }
            sum_slot = a;
        } else if (plan->aggs[a]->func != YT_AGG_SUM1) {
            set_err(errbuf, errlen, "merge_str: sum/sum(1) only");
            return YT_ERR_UNSUPPORTED;
        }
    }
    int sum_is_double = 0;
    if (sum_slot >= 0) {
        uint8_t ct2[kMaxCols];
        memset(ct2, YT_VT_INT64, sizeof(ct2));
        if (col_types) {
            for (int c2 = 0; c2 < kMaxCols; c2++) ct2[c2] = col_types[c2];
        }
        sum_is_double =
            expr_static_type(plan->aggs[sum_slot]->arg, ct2) == YT_VT_DOUBLE;
    }
    YtExecOptions defopt;
    memset(&defopt, 0, sizeof(defopt));
    if (!options) options = &defopt;
    if (stats) memset(stats, 0, sizeof(*stats));
    hipStream_t st = (hipStream_t)(uintptr_t)options->stream;
    double tw0 = now_ms();

    int64_t n = 0, pool_total = 0;
    std::vector<int64_t> row_base(nseg_in);
    std::vector<unsigned long long> pool_base(nseg_in);
    for (int s = 0; s < nseg_in; s++) {
        row_base[s] = n;
        pool_base[s] = (unsigned long long)pool_total;
        n += seg_counts[s];
        pool_total += seg_pool_bytes[s];
    }
    output->row_count = 0;
    output->string_pool_used = 0;
    output->column_count = 1 + plan->agg_count;
    if (n == 0) return YT_OK;

    int64_t* d_rowb = nullptr;
    unsigned long long* d_poolb = nullptr;
    uint64_t* d_hashes = nullptr;
    uint64_t* d_idents = nullptr;
    ulonglong2* d_pfxs = nullptr;
    uint64_t* d_absoff = nullptr;
    StrSlot* d_slots = nullptr;
    TableHdr* d_th = nullptr;
    OutStrState* d_out = nullptr;
    char* d_opool = nullptr;
    unsigned long long* d_ctr = nullptr;
    OutStrState* hgroups = nullptr;
    char* hpool = nullptr;
    auto putall = [&]() {
        g_pool.put(d_rowb); g_pool.put(d_poolb); g_pool.put(d_hashes);
        g_pool.put(d_idents); g_pool.put(d_pfxs); g_pool.put(d_absoff);
        g_pool.put(d_slots); g_pool.put(d_th); g_pool.put(d_out);
        g_pool.put(d_opool); g_pool.put(d_ctr); g_pool.put(hgroups);
        g_pool.put(hpool);
    };
    {
    HIP_CHECK(pool_alloc(&d_rowb, sizeof(int64_t) * nseg_in));
    HIP_CHECK(pool_alloc(&d_poolb, sizeof(unsigned long long) * nseg_in));
    HIP_CHECK(hipMemcpyAsync(d_rowb, row_base.data(),
                             sizeof(int64_t) * nseg_in,
                             hipMemcpyHostToDevice, st));
    HIP_CHECK(hipMemcpyAsync(d_poolb, pool_base.data(),
                             sizeof(unsigned long long) * nseg_in,
                             hipMemcpyHostToDevice, st));
    HIP_CHECK(pool_alloc(&d_hashes, sizeof(uint64_t) * n));
    HIP_CHECK(pool_alloc(&d_idents, sizeof(uint64_t) * n));
    HIP_CHECK(pool_alloc(&d_pfxs, sizeof(ulonglong2) * n));
    HIP_CHECK(pool_alloc(&d_absoff, sizeof(uint64_t) * n));
    uint64_t nslots_cap = next_pow2((uint64_t)n * 2);
    if (nslots_cap < 2048) nslots_cap = 2048;
    uint64_t nslots = options->max_groups_hint > 0
        ? next_pow2((uint64_t)options->max_groups_hint * 2)
        : nslots_cap;
    if (nslots < 2048) nslots = 2048;
    if (nslots > nslots_cap) nslots = nslots_cap;
    HIP_CHECK(pool_alloc(&d_th, sizeof(TableHdr)));
    HIP_CHECK(ytql_launch_strst_hash((const YtStateRow*)states_device, n,
                                     d_rowb, d_poolb, nseg_in,
                                     (const char*)pool_device,
                                     d_hashes, d_idents, d_pfxs, d_absoff, st));
    TableHdr th;
    for (;;) {
        HIP_CHECK(pool_alloc(&d_slots, sizeof(StrSlot) * nslots));
        HIP_CHECK(hipMemsetAsync(d_slots, 0, sizeof(StrSlot) * nslots, st));
        memset(&th, 0, sizeof(th));
        th.group_limit = options->group_row_limit;
        HIP_CHECK(hipMemcpyAsync(d_th, &th, sizeof(th),
                                 hipMemcpyHostToDevice, st));
        HIP_CHECK(ytql_launch_strst_merge((const YtStateRow*)states_device, n,
                                          (const char*)pool_device, d_hashes,
                                          d_idents, d_pfxs, d_absoff, sum_slot,
                                          d_slots, nslots, d_th, st));
        HIP_CHECK(hipMemcpy(&th, d_th, sizeof(th), hipMemcpyDeviceToHost));
        if (th.overflow != 1 || nslots >= nslots_cap) break;
        g_pool.put(d_slots);
        d_slots = nullptr;
        nslots *= 4;
        if (nslots > nslots_cap) nslots = nslots_cap;
    }
    if (th.overflow == 1) {
        putall();
        set_err(errbuf, errlen, "merge_str: table overflow");
        return YT_ERR_CAPACITY;
    }
    int64_t ngroups = (int64_t)th.ngroups;
    int has_null = th.side_used[1] ? 1 : 0;
    if (ngroups + has_null > output->capacity_rows) {
        putall();
        set_err(errbuf, errlen, "merge_str: output rowset too small");
        return YT_ERR_CAPACITY;
    }
    uint64_t opool_cap = (uint64_t)(pool_total ? pool_total : 1);
    HIP_CHECK(pool_alloc(&d_out, sizeof(OutStrState) * (ngroups ? ngroups : 1)));
    HIP_CHECK(pool_alloc(&d_opool, opool_cap));
    HIP_CHECK(pool_alloc(&d_ctr, 2 * sizeof(unsigned long long)));
    HIP_CHECK(hipMemsetAsync(d_ctr, 0, 2 * sizeof(unsigned long long), st));
    if (ngroups > 0) {
        HIP_CHECK(ytql_launch_strst_compact(d_slots, nslots, d_absoff,
                                            (const char*)pool_device, d_out,
                                            d_ctr, d_opool, d_ctr + 1,
                                            opool_cap, d_th, st));
    }
    unsigned long long hctr[2] = {0, 0};
    HIP_CHECK(pool_alloc_host(&hgroups,
                              sizeof(OutStrState) * (ngroups ? ngroups : 1)));
    if (ngroups > 0) {
        HIP_CHECK(hipMemcpyAsync(hgroups, d_out,
                                 sizeof(OutStrState) * ngroups,
                                 hipMemcpyDeviceToHost, st));
    }
    HIP_CHECK(hipMemcpyAsync(hctr, d_ctr, 2 * sizeof(unsigned long long),
                             hipMemcpyDeviceToHost, st));
    HIP_CHECK(hipStreamSynchronize(st));
    if ((unsigned long long)hctr[1] > (unsigned long long)output->string_pool_capacity
        || (output->string_pool == nullptr && hctr[1] != 0)) {
        putall();
        set_err(errbuf, errlen, "merge_str: string pool too small");
        return YT_ERR_CAPACITY;
    }
    if (hctr[1]) {
        HIP_CHECK(pool_alloc_host(&hpool, hctr[1]));
        HIP_CHECK(hipMemcpyAsync(hpool, d_opool, hctr[1],
                                 hipMemcpyDeviceToHost, st));
        HIP_CHECK(hipStreamSynchronize(st));
        size_t total_b = (size_t)hctr[1];
        int ct = (int)std::min<size_t>(std::thread::hardware_concurrency(),
                                       (total_b + (64 << 20) - 1) / (64 << 20));
        if (ct > 1) {
            std::vector<std::thread> cth;
            size_t per = (total_b + ct - 1) / ct;
            for (int t2 = 0; t2 < ct; t2++) {
                size_t b = (size_t)t2 * per;
                size_t e = std::min(total_b, b + per);
                if (b >= e) break;
                cth.emplace_back([&, b, e] {
                    memcpy(output->string_pool + b, hpool + b, e - b);
                });
            }
            for (auto& th2 : cth) th2.join();
        } else {
            memcpy(output->string_pool, hpool, total_b);
        }
    }
    output->string_pool_used = hctr[1];
    int ncols = 1 + plan->agg_count;
    int agg_is_sum1[kMaxAggs];
    for (int a = 0; a < plan->agg_count; a++)
        agg_is_sum1[a] = plan->aggs[a]->func == YT_AGG_SUM1;
    auto emit_range = [&](int64_t b, int64_t e) {
        for (int64_t g = b; g < e; g++) {
            const OutStrState& gg = hgroups[g];
            YtValue* dst = output->values + g * ncols;
            dst[0].id = 0;
            dst[0].flags = 0;
            dst[0].type = YT_VT_STRING;
            dst[0].length = (uint32_t)(gg.off_len & 0xFFFFFF);
            dst[0].data.str = output->string_pool + (gg.off_len >> 24);
            for (int a = 0; a < plan->agg_count; a++) {
                YtValue& v = dst[1 + a];
                v.id = (uint16_t)(1 + a);
                v.flags = 0;
                v.length = 0;
                if (agg_is_sum1[a]) {
                    v.type = YT_VT_INT64;
                    v.data.bits = gg.cnt;
                } else if (gg.nonnull == 0) {
                    v.type = YT_VT_NULL;
                    v.data.bits = 0;
                } else {
                    v.type = sum_is_double ? YT_VT_DOUBLE : YT_VT_INT64;
                    v.data.bits = gg.sum_bits;
                }
            }
        }
    };
    if (ngroups >= 65536) {
        int nt = (int)std::min<int64_t>(std::thread::hardware_concurrency(),
                                        (ngroups + 65535) / 65536);
        std::vector<std::thread> ths;
        int64_t per = (ngroups + nt - 1) / nt;
        for (int t2 = 0; t2 < nt; t2++) {
            int64_t b = (int64_t)t2 * per;
            int64_t e = std::min<int64_t>(ngroups, b + per);
            if (b >= e) break;
            ths.emplace_back(emit_range, b, e);
        }
        for (auto& th2 : ths) th2.join();
    } else {
        emit_range(0, ngroups);
    }
    output->row_count = ngroups;
    if (has_null) {
        YtValue* dst = output->values + output->row_count * ncols;
        uint64_t cnt = th.side_cnt[1];
        uint64_t sum_bits = sum_slot >= 0 ? th.side_agg[1][2 * sum_slot] : 0;
        uint64_t nonnull = sum_slot >= 0 ? th.side_agg[1][2 * sum_slot + 1] : 0;
        dst[0].id = 0;
        dst[0].flags = 0;
        dst[0].type = YT_VT_NULL;
        dst[0].length = 0;
        dst[0].data.bits = 0;
        for (int a = 0; a < plan->agg_count; a++) {
            YtValue& v = dst[1 + a];
            v.id = (uint16_t)(1 + a);
            v.flags = 0;
            v.length = 0;
            if (agg_is_sum1[a]) {
                v.type = YT_VT_INT64;
                v.data.bits = cnt;
            } else if (nonnull == 0) {
                v.type = YT_VT_NULL;
                v.data.bits = 0;
            } else {
                v.type = sum_is_double ? YT_VT_DOUBLE : YT_VT_INT64;
                v.data.bits = sum_bits;
            }
        }
        output->row_count++;
    }
    if (stats) {
        stats->rows_written = output->row_count;
        stats->grouped_row_count = ngroups + has_null;
        stats->incomplete_output = (th.overflow == 2);
        stats->execute_time_ms = now_ms() - tw0;
    }
    putall();
    return YT_OK;
    }
fail:
    putall();
    return rc;
}
