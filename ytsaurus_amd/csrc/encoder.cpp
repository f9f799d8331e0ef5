/* encoder.cpp — host-side chunk writer: produces byte-identical unversioned
 * column segments to the reference's writers. This is the synthetic-data
 * generator for tests and benchmarks (never timed).
 *
 * Restates (/root/reference, ytsaurus/ytsaurus @2026-08-21):
 *   - bit-pack:          core/misc/bit_packed_unsigned_vector-inl.h:15-82
 *                        (header = count(56b)|width(8b); width = MSB+1;
 *                         zeroed dst; width-64 raw path)
 *   - int64 writer:      ytlib/table_chunk_format/integer_column_writer.cpp
 *                        (zigzag encode :24-27; statistics :60-66; direct dump
 *                         :68-83; dictionary dump :85-112 with first-appearance
 *                         ids, 0 = null; RLE dumps :394-489; segment-size
 *                         estimation + type choice :353-391,491-535; enum order
 *                         private.h:25-30 DictionaryRle=0, DictionaryDense=1,
 *                         DirectRle=2, DirectDense=3, first minimum wins;
 *                         segment split at MaxValueCount :565-571;
 *                         RowCount_ in the size estimate is the CUMULATIVE
 *                         column row count at dump time)
 *   - null bitmap:       core/misc/bitmap.h TBitmapOutput (ui8 LSB-first,
 *                        1 = null, serialized size aligned to 8, zero padding)
 *   - double writer:     ytlib/table_chunk_format/floating_point_column_writer.cpp
 *                        ([ui64 count][raw doubles][null bitmap])
 */
#include <stdint.h>
#include <string.h>
#include <stdio.h>
#include <stdlib.h>
#include <vector>
#include <unordered_map>

#include "../../include/ytql_gpu.h"

static void set_err(char* errbuf, size_t errlen, const char* msg)
{
    if (errbuf && errlen) snprintf(errbuf, errlen, "%s", msg);
}

static inline uint64_t zigzag_encode64(int64_t n)
{
    return ((uint64_t)n << 1) ^ (uint64_t)(n >> 63);
}

/* GetWidth — bit_packed_unsigned_vector-inl.h:15-18 */
static inline uint64_t bp_width(uint64_t value)
{
    if (value == 0) return 0;
    return 64 - __builtin_clzll(value);
}

extern "C" int64_t yt_bitpack_size_words(uint64_t max_value, int64_t n)
{
    return 1 + (int64_t)((bp_width(max_value) * (uint64_t)n + 63ULL) >> 6);
}

/* BitPackUnsignedVector — -inl.h:32-82. dst must be zeroed. */
extern "C" int64_t yt_bitpack(const uint64_t* values, int64_t n, uint64_t max_value,
                              uint64_t* dst)
{
    uint64_t width = bp_width(max_value);
    uint64_t header = (uint64_t)n;
    header |= width << 56;
    *dst = header;
    if (max_value == 0) {
        return 1;
    }
    uint64_t* word = dst + 1;
    uint8_t offset = 0;
    if (width < 64) {
        for (int64_t i = 0; i < n; i++) {
            uint64_t x = values[i];
            if (offset + width < 64) {
                *word |= (x << offset);
                offset += width;
            } else {
                *word |= (x << offset);
                offset = (uint8_t)(offset + width);
                offset &= 0x3F;
                ++word;
                x >>= width - offset;
                if (x > 0) {
                    *word |= x;
                }
            }
        }
    } else {
        for (int64_t i = 0; i < n; i++) {
            *word = values[i];
            ++word;
        }
    }
    return (offset == 0 ? 0 : 1) + (word - dst);
}

/* CompressedUnsignedVectorSizeInBytes — -inl.h:20-30 */
static inline int64_t cs_bytes(uint64_t max_value, int64_t count)
{
    return yt_bitpack_size_words(max_value, count) * 8;
}

/* ------------------------------------------------------------------ */

namespace {

struct Blob {
    std::vector<uint8_t> bytes;

    int64_t size() const { return (int64_t)bytes.size(); }

    /* append a bit-packed vector */
    void pack(const std::vector<uint64_t>& v, uint64_t max_value)
    {
        int64_t words = yt_bitpack_size_words(max_value, (int64_t)v.size());
        size_t at = bytes.size();
        bytes.resize(at + words * 8, 0);
        yt_bitpack(v.data(), (int64_t)v.size(), max_value, (uint64_t*)(bytes.data() + at));
    }

    /* append a null bitmap: 1 bit per entry, LSB-first, aligned 8, zero pad */
    void bitmap(const std::vector<uint8_t>& isnull)
    {
        size_t nbytes = (isnull.size() + 7) / 8;
        nbytes = (nbytes + 7) & ~(size_t)7;
        size_t at = bytes.size();
        bytes.resize(at + nbytes, 0);
        for (size_t i = 0; i < isnull.size(); i++) {
            if (isnull[i]) bytes[at + i / 8] |= (uint8_t)(1u << (i % 8));
        }
    }

    void raw(const void* p, size_t n)
    {
        size_t at = bytes.size();
        bytes.resize(at + n);
        memcpy(bytes.data() + at, p, n);
    }
};

struct SegmentOut {
    int32_t type;
    int32_t row_count;
    uint64_t min_value;
    Blob blob;
};

/* One integer segment: values already zigzag/raw-encoded ("data" space),
 * nulls as bytemask. cum_row_count = column rows INCLUDING this segment
 * (the reference's RowCount_ at dump time, used only in size estimation). */
static void dump_int_segment(const uint64_t* data, const uint8_t* isnull, int64_t n,
                             int64_t cum_row_count, SegmentOut* out)
{
    /* statistics — integer_column_writer.cpp:60-66 + AddValues :573-600 */
    uint64_t minv = UINT64_MAX, maxv = 0;
    std::unordered_map<uint64_t, int> distinct;   /* value -> first-appearance id (1-based) */
    std::vector<uint64_t> first_order;            /* values in first-appearance order */
    int64_t run_count = 0;
    for (int64_t i = 0; i < n; i++) {
        if (!isnull[i]) {
            uint64_t v = data[i];
            if (v < minv) minv = v;
            if (v > maxv) maxv = v;
            auto it = distinct.emplace(v, (int)distinct.size() + 1);
            if (it.second) first_order.push_back(v);
        }
        if (i == 0 || isnull[i] != isnull[i - 1] || data[i] != data[i - 1]) {
            run_count++;
        }
    }
    int64_t ndistinct = (int64_t)distinct.size();
    uint64_t span = maxv - minv;   /* wraps to 1 for all-null segments, as in the reference */

    /* segment-size estimation — integer_column_writer.cpp:353-391 */
    int64_t sz[4];
    sz[YT_SEG_DICTIONARY_RLE] = cs_bytes(span, ndistinct)
                              + cs_bytes((uint64_t)ndistinct + 1, run_count)
                              + cs_bytes((uint64_t)cum_row_count, run_count);
    sz[YT_SEG_DICTIONARY_DENSE] = cs_bytes(span, ndistinct)
                                + cs_bytes((uint64_t)ndistinct + 1, n);
    sz[YT_SEG_DIRECT_RLE] = cs_bytes(span, run_count)
                          + cs_bytes((uint64_t)cum_row_count, run_count)
                          + run_count / 8;
    sz[YT_SEG_DIRECT_DENSE] = cs_bytes(span, n)
                            + n / 8;
    int best = 0;
    for (int t = 1; t < 4; t++) {
        if (sz[t] < sz[best]) best = t;   /* first minimum in enum order wins */
    }

    out->type = best;
    out->row_count = (int32_t)n;
    out->min_value = minv;

    switch (best) {
    case YT_SEG_DIRECT_DENSE: {
        /* DumpDirectValues :68-83: subtract min from non-null entries */
        std::vector<uint64_t> vals(n);
        std::vector<uint8_t> nb(isnull, isnull + n);
        for (int64_t i = 0; i < n; i++) {
            vals[i] = isnull[i] ? data[i] : data[i] - minv;
        }
        out->blob.pack(vals, span);
        out->blob.bitmap(nb);
        break;
    }
    case YT_SEG_DICTIONARY_DENSE: {
        /* DumpDictionaryValues :85-112 */
        std::vector<uint64_t> dict;
        dict.reserve(ndistinct);
        for (uint64_t v : first_order) dict.push_back(v - minv);
        std::vector<uint64_t> ids(n);
        for (int64_t i = 0; i < n; i++) {
            ids[i] = isnull[i] ? 0 : (uint64_t)distinct[data[i]];
        }
        out->blob.pack(dict, span);
        out->blob.pack(ids, (uint64_t)ndistinct + 1);
        break;
    }
    case YT_SEG_DIRECT_RLE: {
        /* DumpDirectRleValues :394-437 */
        std::vector<uint64_t> run_vals;
        std::vector<uint8_t> run_null;
        std::vector<uint64_t> run_starts;
        for (int64_t i = 0; i < n;) {
            int64_t j = i + 1;
            while (j < n && data[i] == data[j] && isnull[i] == isnull[j]) j++;
            run_vals.push_back(isnull[i] ? 0 : data[i] - minv);
            run_null.push_back(isnull[i]);
            run_starts.push_back((uint64_t)i);
            i = j;
        }
        out->blob.pack(run_vals, span);
        out->blob.bitmap(run_null);
        out->blob.pack(run_starts, run_starts.back());
        break;
    }
    case YT_SEG_DICTIONARY_RLE: {
        /* DumpDictionaryRleValues :439-489 */
        std::vector<uint64_t> dict;
        dict.reserve(ndistinct);
        for (uint64_t v : first_order) dict.push_back(v - minv);
        std::vector<uint64_t> run_ids;
        std::vector<uint64_t> run_starts;
        for (int64_t i = 0; i < n;) {
            int64_t j = i + 1;
            while (j < n && data[i] == data[j] && isnull[i] == isnull[j]) j++;
            run_ids.push_back(isnull[i] ? 0 : (uint64_t)distinct[data[i]]);
            run_starts.push_back((uint64_t)i);
            i = j;
        }
        out->blob.pack(dict, span);
        out->blob.pack(run_ids, (uint64_t)ndistinct + 1);
        out->blob.pack(run_starts, run_starts.back());
        break;
    }
    }
}

static int finish_column(std::vector<SegmentOut>& segs, YtEncodedColumn* out,
                         char* errbuf, size_t errlen)
{
    int64_t total = 0;
    for (auto& s : segs) total += s.blob.size();
    int64_t meta_bytes = (int64_t)(sizeof(YtSegment) * segs.size());
    uint8_t* blob = (uint8_t*)malloc(meta_bytes + total);
    if (!blob) { set_err(errbuf, errlen, "oom"); return YT_ERR_CAPACITY; }
    YtSegment* metas = (YtSegment*)blob;
    uint8_t* p = blob + meta_bytes;
    for (size_t i = 0; i < segs.size(); i++) {
        metas[i].type = segs[i].type;
        metas[i].row_count = segs[i].row_count;
        metas[i].min_value = segs[i].min_value;
        metas[i].data = p;
        metas[i].data_size = segs[i].blob.size();
        memcpy(p, segs[i].blob.bytes.data(), segs[i].blob.size());
        p += segs[i].blob.size();
    }
    out->segment_count = (int32_t)segs.size();
    out->segments = metas;
    out->blob = blob;
    out->blob_size = meta_bytes + total;
    return YT_OK;
}

} /* namespace */

extern "C" int yt_encode_int64_column(
    const int64_t* values, const uint8_t* nulls, int64_t n,
    int32_t max_segment_values, int32_t is_unsigned, int64_t cum_rows_base,
    YtEncodedColumn* out, char* errbuf, size_t errlen)
{
    if (max_segment_values <= 0) max_segment_values = 128 * 1024;  /* DefaultMaxSegmentValueCount */
    std::vector<SegmentOut> segs;
    std::vector<uint64_t> data;
    std::vector<uint8_t> isnull;
    int64_t cum = cum_rows_base;
    for (int64_t i = 0; i < n; i++) {
        int nu = nulls ? nulls[i] : 0;
        /* null values store data 0 — AddValues :578-584 */
        uint64_t d = 0;
        if (!nu) d = is_unsigned ? (uint64_t)values[i] : zigzag_encode64(values[i]);
        data.push_back(d);
        isnull.push_back((uint8_t)nu);
        cum++;
        if ((int64_t)data.size() >= max_segment_values) {
            segs.emplace_back();
            dump_int_segment(data.data(), isnull.data(), (int64_t)data.size(), cum,
                             &segs.back());
            data.clear();
            isnull.clear();
        }
    }
    if (!data.empty()) {
        segs.emplace_back();
        dump_int_segment(data.data(), isnull.data(), (int64_t)data.size(), cum,
                         &segs.back());
    }
    return finish_column(segs, out, errbuf, errlen);
}

extern "C" int yt_encode_double_column(
    const double* values, const uint8_t* nulls, int64_t n,
    int32_t max_segment_values,
    YtEncodedColumn* out, char* errbuf, size_t errlen)
{
    if (max_segment_values <= 0) max_segment_values = 128 * 1024;
    std::vector<SegmentOut> segs;
    int64_t at = 0;
    while (at < n || (n == 0 && segs.empty() && at == 0 && n > 0)) {
        int64_t cnt = n - at < max_segment_values ? n - at : max_segment_values;
        if (cnt <= 0) break;
        segs.emplace_back();
        SegmentOut& s = segs.back();
        s.type = YT_SEG_DOUBLE;
        s.row_count = (int32_t)cnt;
        s.min_value = 0;
        uint64_t c64 = (uint64_t)cnt;
        s.blob.raw(&c64, 8);
        s.blob.raw(values + at, 8 * cnt);
        std::vector<uint8_t> nb(cnt, 0);
        if (nulls) for (int64_t i = 0; i < cnt; i++) nb[i] = nulls[at + i];
        s.blob.bitmap(nb);
        at += cnt;
    }
    return finish_column(segs, out, errbuf, errlen);
}

extern "C" void yt_encoded_column_free(YtEncodedColumn* col)
{
    if (col && col->blob) {
        free(col->blob);
        col->blob = nullptr;
        col->segments = nullptr;
        col->segment_count = 0;
    }
}
