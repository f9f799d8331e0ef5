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
#include <unordered_set>
#include <string_view>
#include <string>

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

/* boolean_column_writer.cpp:17-28 DumpBooleanValues */
extern "C" int yt_encode_bool_column(
    const uint8_t* values, const uint8_t* nulls, int64_t n,
    int32_t max_segment_values,
    YtEncodedColumn* out, char* errbuf, size_t errlen)
{
    if (max_segment_values <= 0) max_segment_values = 128 * 1024;
    std::vector<SegmentOut> segs;
    int64_t at = 0;
    while (at < n) {
        int64_t cnt = n - at < max_segment_values ? n - at : max_segment_values;
        segs.emplace_back();
        SegmentOut& s = segs.back();
        s.type = YT_SEG_BOOLEAN;
        s.row_count = (int32_t)cnt;
        s.min_value = 0;
        uint64_t c64 = (uint64_t)cnt;
        s.blob.raw(&c64, 8);
        std::vector<uint8_t> vb(cnt), nb(cnt, 0);
        for (int64_t i = 0; i < cnt; i++) {
            vb[i] = values[at + i] ? 1 : 0;
            if (nulls) nb[i] = nulls[at + i];
        }
        s.blob.bitmap(vb);     /* value bitmap (bit set = true) */
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

/* ------------------------------------------------------------------ */
/* string columns — string_column_writer.cpp (unversioned):
 *   CaptureValue :100-151 (dictionary ids in first-appearance order, 1-based,
 *   empty-vs-null via pointer sentinel), DumpDictionaryValues :153-202,
 *   DumpDirectValues :205-231, RLE dumps :497-594, segment choice :592-648,
 *   size estimates :651-680; offsets stored as ZigZagEncode32 of the diff
 *   from an expected running length (PrepareDiffFromExpected,
 *   core/misc/bit_packed_unsigned_vector.cpp:11-31, expected =
 *   DivRound(last, count), numeric_helpers-inl.h:29-33).
 * YtSegment.min_value carries TStringSegmentMeta.expected_length. */

static inline uint32_t zigzag_encode32(int32_t n)
{
    return ((uint32_t)n << 1) ^ (uint32_t)(n >> 31);
}

static uint32_t prepare_diff_from_expected(std::vector<uint64_t>* values,
                                           uint64_t* max_diff)
{
    *max_diff = 0;
    if (values->empty()) return 0;
    int64_t last = (int64_t)values->back();
    int64_t cnt = (int64_t)values->size();
    lldiv_t d = lldiv(last, cnt);
    uint32_t expected = (uint32_t)(d.quot + (d.rem >= (cnt + 1) / 2 ? 1 : 0));
    int64_t expected_value = 0;
    for (size_t i = 0; i < values->size(); i++) {
        expected_value += expected;
        int32_t diff = (int32_t)((int64_t)(*values)[i] - expected_value);
        (*values)[i] = zigzag_encode32(diff);
        if ((*values)[i] > *max_diff) *max_diff = (*values)[i];
    }
    return expected;
}

namespace {

struct StrSeg {
    /* per-segment string state (Values_ as (off,len,null) views) */
    std::vector<uint64_t> begins;
    std::vector<uint32_t> lens;
    std::vector<uint8_t> nulls;
    std::unordered_map<std::string_view, uint32_t> dict;  /* first-appearance ids */
    uint64_t dict_bytes = 0;
    uint32_t max_len = 0;
    int64_t direct_bytes = 0;      /* Σ lens (DirectBuffer_ size) */
    int64_t rle_bytes = 0;         /* Σ run-head lens */
    std::vector<uint64_t> rle_starts;
};

static void dump_string_segment(const char* blob, StrSeg& st, int64_t cum_rows,
                                SegmentOut* out)
{
    const int64_t n = (int64_t)st.begins.size();
    const int64_t run_count = (int64_t)st.rle_starts.size();
    const int64_t ndict = (int64_t)st.dict.size();
    (void)cum_rows;

    auto sv = [&](int64_t i) -> std::string_view {
        return std::string_view(blob + st.begins[i], st.lens[i]);
    };

    /* GetSegmentSize :651-680; enum order DictionaryRle=0, DictionaryDense=1,
     * DirectRle=2, DirectDense=3 (private.h:33-38), first minimum wins */
    int64_t sz[4];
    sz[YT_SEG_DICTIONARY_RLE] = st.dict_bytes
        + cs_bytes(st.max_len, ndict)
        + cs_bytes((uint64_t)ndict + 1, run_count)
        + cs_bytes((uint64_t)n, run_count);
    sz[YT_SEG_DICTIONARY_DENSE] = st.dict_bytes
        + cs_bytes(st.max_len, ndict)
        + cs_bytes((uint64_t)ndict + 1, n);
    sz[YT_SEG_DIRECT_RLE] = st.rle_bytes
        + cs_bytes(st.max_len, run_count)
        + cs_bytes((uint64_t)n, run_count)
        + n / 8;
    sz[YT_SEG_DIRECT_DENSE] = st.direct_bytes
        + cs_bytes(st.max_len, n)
        + n / 8;
    int best = 0;
    for (int t = 1; t < 4; t++) {
        if (sz[t] < sz[best]) best = t;
    }

    out->type = best;
    out->row_count = (int32_t)n;

    uint64_t expected = 0;
    switch (best) {
    case YT_SEG_DIRECT_DENSE: {
        /* [packed zigzag-diff end-offsets][null bitmap][string data] */
        std::vector<uint64_t> offsets(n);
        uint64_t run = 0;
        for (int64_t i = 0; i < n; i++) {
            run += st.lens[i];
            offsets[i] = run;
        }
        uint64_t maxd;
        expected = prepare_diff_from_expected(&offsets, &maxd);
        out->blob.pack(offsets, maxd);
        out->blob.bitmap(st.nulls);
        for (int64_t i = 0; i < n; i++) {
            if (!st.nulls[i]) out->blob.raw(blob + st.begins[i], st.lens[i]);
        }
        break;
    }
    case YT_SEG_DICTIONARY_DENSE: {
        /* [packed ids][packed zigzag-diff dict end-offsets][dict data] */
        std::vector<uint64_t> ids(n);
        std::vector<uint64_t> dict_offsets;
        std::vector<int64_t> dict_rows;   /* row index of each dict entry's first use */
        uint64_t doff = 0;
        uint32_t dsize = 0;
        for (int64_t i = 0; i < n; i++) {
            if (st.nulls[i]) { ids[i] = 0; continue; }
            uint32_t id = st.dict[sv(i)];
            ids[i] = id;
            if (id > dsize) {
                doff += st.lens[i];
                dict_offsets.push_back(doff);
                dict_rows.push_back(i);
                dsize++;
            }
        }
        out->blob.pack(ids, (uint64_t)dsize + 1);
        uint64_t maxd;
        expected = prepare_diff_from_expected(&dict_offsets, &maxd);
        out->blob.pack(dict_offsets, maxd);
        for (int64_t r : dict_rows) out->blob.raw(blob + st.begins[r], st.lens[r]);
        break;
    }
    case YT_SEG_DIRECT_RLE: {
        /* [packed run starts][packed zigzag-diff run end-offsets]
         * [run null bitmap][run string data] */
        std::vector<uint64_t> offsets;
        std::vector<uint8_t> run_null;
        uint64_t run = 0;
        for (uint64_t ri : st.rle_starts) {
            run += st.lens[ri];
            offsets.push_back(run);
            run_null.push_back(st.nulls[ri]);
        }
        out->blob.pack(st.rle_starts, st.rle_starts.back());
        uint64_t maxd;
        expected = prepare_diff_from_expected(&offsets, &maxd);
        out->blob.pack(offsets, maxd);
        out->blob.bitmap(run_null);
        for (uint64_t ri : st.rle_starts) {
            if (!st.nulls[ri]) out->blob.raw(blob + st.begins[ri], st.lens[ri]);
        }
        break;
    }
    case YT_SEG_DICTIONARY_RLE: {
        /* [packed run starts][packed run ids][packed zigzag-diff dict
         * end-offsets][dict data] — NB ids maxValue = dict.size() here
         * (DumpDictionaryRleData :578), unlike the dense variant's size+1 */
        std::vector<uint64_t> ids;
        std::vector<uint64_t> dict_offsets;
        std::vector<int64_t> dict_rows;
        uint64_t doff = 0;
        uint32_t dsize = 0;
        for (uint64_t ri : st.rle_starts) {
            if (st.nulls[ri]) { ids.push_back(0); continue; }
            uint32_t id = st.dict[sv(ri)];
            ids.push_back(id);
            if (id > dsize) {
                doff += st.lens[ri];
                dict_offsets.push_back(doff);
                dict_rows.push_back((int64_t)ri);
                dsize++;
            }
        }
        out->blob.pack(st.rle_starts, st.rle_starts.back());
        out->blob.pack(ids, (uint64_t)st.dict.size());
        uint64_t maxd;
        expected = prepare_diff_from_expected(&dict_offsets, &maxd);
        out->blob.pack(dict_offsets, maxd);
        for (int64_t r : dict_rows) out->blob.raw(blob + st.begins[r], st.lens[r]);
        break;
    }
    }
    out->min_value = expected;    /* carries expected_length for strings */
}

} /* namespace */

extern "C" int yt_encode_string_column(
    const char* blob, const uint64_t* begins, const uint32_t* lens,
    const uint8_t* nulls, int64_t n, int32_t max_segment_values,
    YtEncodedColumn* out, char* errbuf, size_t errlen)
{
    if (max_segment_values <= 0) max_segment_values = 128 * 1024;
    std::vector<SegmentOut> segs;
    StrSeg st;
    int64_t cum = 0;
    auto flush = [&]() {
        if (st.begins.empty()) return;
        segs.emplace_back();
        dump_string_segment(blob, st, cum, &segs.back());
        st = StrSeg();
    };
    for (int64_t i = 0; i < n; i++) {
        int nu = nulls ? nulls[i] : 0;
        uint64_t b = nu ? 0 : begins[i];
        uint32_t l = nu ? 0 : lens[i];
        /* RLE run break — AreValuesEqual :233-247 (null==null; bytes equal) */
        bool same = false;
        if (!st.begins.empty()) {
            size_t k = st.begins.size() - 1;
            if (nu && st.nulls[k]) same = true;
            else if (!nu && !st.nulls[k] && st.lens[k] == l &&
                     memcmp(blob + st.begins[k], blob + b, l) == 0) same = true;
        }
        if (!same) {
            st.rle_starts.push_back(st.begins.size());
            st.rle_bytes += l;
        }
        if (!nu) {
            auto ins = st.dict.emplace(std::string_view(blob + b, l),
                                       (uint32_t)st.dict.size() + 1);
            if (ins.second) {
                st.dict_bytes += l;
                if (l > st.max_len) st.max_len = l;
            }
            st.direct_bytes += l;
        }
        st.begins.push_back(b);
        st.lens.push_back(l);
        st.nulls.push_back((uint8_t)nu);
        cum++;
        if ((int64_t)st.begins.size() >= max_segment_values ||
            st.direct_bytes > (int64_t)32 * 1024 * 1024) {
            flush();
        }
    }
    flush();
    return finish_column(segs, out, errbuf, errlen);
}

/* ------------------------------------------------------------------ */
/* versioned scan-format slice — see include/ytql_gpu.h
 * for the layout citations (timestamp_writer.cpp DumpSegment,
 * column_writer_detail.cpp DumpVersionedData,
 * integer_column_writer.cpp:119-246 DumpDirectValues). */

/* shared implementation for versioned int64 / double columns.
 * Layout per segment (column_writer_detail.cpp DumpVersionedData +
 * integer/floating_point versioned DumpSegment):
 *   [value index: dense = cumulative values-per-row diff-from-expected |
 *                 sparse = row index per value]
 *   [per-value timestamp indexes]
 *   [aggregate bitmap]                       (aggregate columns only)
 *   [value part: int direct = values+nullbitmap | int dict = dictionary+ids
 *               | double = u64 count + raw doubles + nullbitmap]
 * Index choice: dense iff denseSize <= sparseSize (DumpVersionedData:201-209).
 * Int value choice: dictionary iff dictionarySize < directSize
 * (integer_column_writer.cpp:222-239). */
static int encode_versioned_impl(
    const uint32_t* writes_per_row, const uint64_t* write_ts,
    const int64_t* ivalues, const double* dvalues,
    const char* sbytes, const uint32_t* slens,
    const uint8_t* value_nulls, const uint8_t* value_agg,
    const uint32_t* deletes_per_row, const uint64_t* delete_ts,
    int64_t row_count, int64_t max_rows_per_segment,
    YtVersionedColumn* out, char* errbuf, size_t errlen)
{
    if (max_rows_per_segment <= 0) max_rows_per_segment = 128 * 1024;
    memset(out, 0, sizeof(*out));
    int nseg = (int)((row_count + max_rows_per_segment - 1) / max_rows_per_segment);
    if (nseg == 0) return YT_OK;
    const int is_double = dvalues != nullptr;
    const int is_str = sbytes != nullptr;

    auto* tsegs = (YtTimestampSeg*)calloc(nseg, sizeof(YtTimestampSeg));
    auto* vsegs = (YtVersionedValueSeg*)calloc(nseg, sizeof(YtVersionedValueSeg));
    auto fail = [&](int rc) {
        for (int k = 0; k < nseg; k++) {
            free((void*)tsegs[k].data);
            free((void*)vsegs[k].data);
        }
        free(tsegs); free(vsegs);
        return rc;
    };

    int64_t wat = 0, dat = 0;   /* global flattened cursors */
    int64_t sbat = 0;           /* global byte cursor (string values) */
    for (int si = 0; si < nseg; si++) {
        int64_t r0 = (int64_t)si * max_rows_per_segment;
        int64_t r1 = r0 + max_rows_per_segment;
        if (r1 > row_count) r1 = row_count;
        int64_t rows = r1 - r0;

        /* --- timestamp segment (timestamp_writer.cpp:45-58,185-196) --- */
        std::vector<uint64_t> dict, wids, dids, wcnt, dcnt;
        std::unordered_map<uint64_t, uint32_t> uniq;
        uint64_t ts_min = ~0ULL, ts_max = 0;
        auto reg = [&](uint64_t ts) -> uint32_t {
            if (ts < ts_min) ts_min = ts;
            if (ts > ts_max) ts_max = ts;
            auto it = uniq.emplace(ts, (uint32_t)dict.size());
            if (it.second) dict.push_back(ts);
            return it.first->second;
        };
        std::vector<uint64_t> vals_zz, tsids, voff;
        std::vector<double> vals_d;
        std::vector<const char*> sptr;          /* string values */
        std::vector<uint32_t> slen2;
        std::vector<uint8_t> vnull, vagg;
        uint64_t vmin = ~0ULL, vmax = 0, max_tsid = 0;
        for (int64_t r = r0; r < r1; r++) {
            uint32_t wc = writes_per_row[r];
            for (uint32_t i = 0; i < wc; i++) {
                uint64_t ts = write_ts[wat + i];
                if (i && write_ts[wat + i - 1] <= ts) {
                    set_err(errbuf, errlen,
                            "versioned: write timestamps must be strictly descending per row");
                    return fail(YT_ERR_INVALID_CHUNK);
                }
                wids.push_back(reg(ts));
                uint8_t nul = value_nulls ? value_nulls[wat + i] : 0;
                if (is_str) {
                    uint32_t sl = slens[wat + i];
                    if (nul && sl != 0) {
                        set_err(errbuf, errlen,
                                "versioned string: null write with nonzero length");
                        return fail(YT_ERR_INVALID_CHUNK);
                    }
                    sptr.push_back(sbytes + sbat);
                    slen2.push_back(sl);
                    sbat += sl;
                } else if (is_double) {
                    vals_d.push_back(nul ? 0.0 : dvalues[wat + i]);
                } else {
                    uint64_t zz = 0;
                    if (!nul) {
                        zz = zigzag_encode64(ivalues[wat + i]);
                        if (zz < vmin) vmin = zz;
                        if (zz > vmax) vmax = zz;
                    }
                    vals_zz.push_back(zz);
                }
                vnull.push_back(nul);
                vagg.push_back(value_agg ? value_agg[wat + i] : 0);
                tsids.push_back(i);
                if (i > max_tsid) max_tsid = i;
            }
            wat += wc;
            wcnt.push_back((wcnt.empty() ? 0 : wcnt.back()) + wc);
            voff.push_back((voff.empty() ? 0 : voff.back()) + wc);
            uint32_t dc = deletes_per_row[r];
            for (uint32_t i = 0; i < dc; i++) {
                uint64_t ts = delete_ts[dat + i];
                if (i && delete_ts[dat + i - 1] <= ts) {
                    set_err(errbuf, errlen,
                            "versioned: delete timestamps must be strictly descending per row");
                    return fail(YT_ERR_INVALID_CHUNK);
                }
                dids.push_back(reg(ts));
            }
            dat += dc;
            dcnt.push_back((dcnt.empty() ? 0 : dcnt.back()) + dc);
        }
        uint64_t ts_span = ts_max - ts_min;   /* wraps when no timestamps */
        std::vector<uint64_t> dict_rel(dict.size());
        for (size_t i = 0; i < dict.size(); i++) dict_rel[i] = dict[i] - ts_min;

        uint64_t wdiff_max = 0, ddiff_max = 0;
        uint32_t exp_w = prepare_diff_from_expected(&wcnt, &wdiff_max);
        uint32_t exp_d = prepare_diff_from_expected(&dcnt, &ddiff_max);

        Blob tb;
        tb.pack(dict_rel, ts_span);
        tb.pack(wids, dict.size());
        tb.pack(dids, dict.size());
        tb.pack(wcnt, wdiff_max);
        tb.pack(dcnt, ddiff_max);

        YtTimestampSeg& T = tsegs[si];
        T.row_count = rows;
        T.base_timestamp = ts_min;
        T.expected_writes_per_row = exp_w;
        T.expected_deletes_per_row = exp_d;
        T.data_size = tb.size();
        void* tp = malloc(tb.size() ? tb.size() : 1);
        memcpy(tp, tb.bytes.data(), tb.size());
        T.data = tp;

        /* --- value segment --- */
        const int64_t nvalues = (int64_t)vnull.size();
        const std::vector<uint64_t> vcum = voff;   /* cumulative, pre-diff */

        /* index layout: dense iff denseSize <= sparseSize
         * (DumpVersionedData:201-209) */
        uint64_t vdiff_max = 0;
        uint32_t exp_v;
        {
            std::vector<uint64_t> probe = voff;
            exp_v = prepare_diff_from_expected(&probe, &vdiff_max);
            voff.swap(probe);   /* voff now holds the diffs */
        }
        int64_t dense_sz = cs_bytes(vdiff_max, rows);
        int64_t sparse_sz = cs_bytes((uint64_t)rows, nvalues);
        int dense = dense_sz <= sparse_sz || nvalues == 0;

        /* string value layout: dictionary iff dictionaryByteSize <
         * directByteSize (string_column_writer.cpp:80-92,338-352); direct =
         * [cumulative END offsets, diff-from-expected][null bitmap][bytes],
         * dictionary = [ids, 0=null first-appearance][cumulative dictionary
         * END offsets, diff-from-expected][dictionary bytes] */
        int use_sdict = 0;
        uint32_t s_expected = 0;
        std::vector<uint64_t> s_offs, s_ids;
        std::vector<char> s_data;
        if (is_str) {
            uint64_t direct_bytes = 0, max_len = 0;
            for (int64_t i = 0; i < nvalues; i++) {
                direct_bytes += slen2[i];
                if (slen2[i] > max_len) max_len = slen2[i];
            }
            std::unordered_map<std::string, uint32_t> suniq;
            std::vector<std::pair<const char*, uint32_t>> sdict;
            uint64_t dict_bytes = 0;
            std::vector<uint64_t> ids1;
            ids1.reserve(nvalues);
            for (int64_t i = 0; i < nvalues; i++) {
                if (vnull[i]) { ids1.push_back(0); continue; }
                auto it = suniq.emplace(std::string(sptr[i], slen2[i]),
                                        (uint32_t)sdict.size() + 1);
                if (it.second) {
                    sdict.push_back({sptr[i], slen2[i]});
                    dict_bytes += slen2[i];
                }
                ids1.push_back(it.first->second);
            }
            int64_t direct_sz = (int64_t)direct_bytes
                              + cs_bytes(max_len, nvalues);
            int64_t dict_sz = (int64_t)dict_bytes
                            + cs_bytes(max_len, (int64_t)sdict.size())
                            + cs_bytes((uint64_t)sdict.size() + 1, nvalues);
            use_sdict = dict_sz < direct_sz;
            uint64_t sdiff_max = 0;
            if (use_sdict) {
                s_ids.swap(ids1);
                uint64_t off = 0;
                for (auto& dv : sdict) {
                    s_data.insert(s_data.end(), dv.first, dv.first + dv.second);
                    off += dv.second;
                    s_offs.push_back(off);
                }
                s_expected = prepare_diff_from_expected(&s_offs, &sdiff_max);
            } else {
                uint64_t off = 0;
                for (int64_t i = 0; i < nvalues; i++) {
                    if (!vnull[i])
                        s_data.insert(s_data.end(), sptr[i], sptr[i] + slen2[i]);
                    off += slen2[i];            /* null steps 0 */
                    s_offs.push_back(off);
                }
                s_expected = prepare_diff_from_expected(&s_offs, &sdiff_max);
            }
            vmax = sdiff_max; vmin = 0;         /* reuse vspan for the pack */
        }

        /* int value layout: dictionary iff dictionarySize < directSize */
        uint64_t vspan = vmax - vmin;         /* wraps when all null/empty */
        int use_dict = 0;
        std::vector<uint64_t> vdict, vids;
        if (!is_double && !is_str) {
            std::unordered_map<uint64_t, uint32_t> vuniq;
            for (int64_t i = 0; i < nvalues; i++) {
                if (!vnull[i]) vuniq.emplace(vals_zz[i], 0);
            }
            int64_t dict_sz = cs_bytes(vspan, (int64_t)vuniq.size())
                            + cs_bytes((uint64_t)vuniq.size() + 1, nvalues);
            int64_t direct_sz = cs_bytes(vspan, nvalues)
                              + (int64_t)((nvalues + 7) / 8);
            use_dict = dict_sz < direct_sz;
            if (use_dict) {
                /* first-appearance order, ids 1-based, 0 = null
                 * (integer_column_writer.cpp DumpDictionaryValues) */
                std::unordered_map<uint64_t, uint32_t> order;
                for (int64_t i = 0; i < nvalues; i++) {
                    if (vnull[i]) { vids.push_back(0); continue; }
                    auto it = order.emplace(vals_zz[i], (uint32_t)vdict.size() + 1);
                    if (it.second) vdict.push_back(vals_zz[i] - vmin);
                    vids.push_back(it.first->second);
                }
            } else {
                for (int64_t i = 0; i < nvalues; i++)
                    if (!vnull[i]) vals_zz[i] -= vmin;
            }
        }

        Blob vb;
        if (dense) {
            vb.pack(voff, vdiff_max);
        } else {
            std::vector<uint64_t> row_idx;
            row_idx.reserve(nvalues);
            for (int64_t r = 0; r < rows; r++) {
                while ((int64_t)row_idx.size() < (int64_t)vcum[r])
                    row_idx.push_back((uint64_t)r);
            }
            vb.pack(row_idx, row_idx.empty() ? 0 : row_idx.back());
        }
        vb.pack(tsids, max_tsid);
        if (value_agg) vb.bitmap(vagg);
        if (is_str) {
            if (use_sdict) {
                vb.pack(s_ids, s_offs.size() + 1);
                vb.pack(s_offs, vspan);         /* vspan = offset diff max */
                vb.raw(s_data.data(), s_data.size());
            } else {
                vb.pack(s_offs, vspan);
                vb.bitmap(vnull);
                vb.raw(s_data.data(), s_data.size());
            }
        } else if (is_double) {
            uint64_t cnt = (uint64_t)nvalues;
            vb.raw(&cnt, 8);
            vb.raw(vals_d.data(), vals_d.size() * 8);
            vb.bitmap(vnull);
        } else if (use_dict) {
            vb.pack(vdict, vspan);
            vb.pack(vids, (uint64_t)vdict.size() + 1);
        } else {
            vb.pack(vals_zz, vspan);
            vb.bitmap(vnull);
        }

        YtVersionedValueSeg& V = vsegs[si];
        V.row_count = rows;
        V.base_value = is_str ? (uint64_t)s_expected : vmin;
        V.expected_values_per_row = exp_v;
        V.type = is_str
            ? (YT_VSEG_STR_DIRECT_DENSE | (use_sdict ? 1 : 0) | (dense ? 0 : 2))
            : is_double
            ? (dense ? YT_VSEG_DOUBLE_DENSE : YT_VSEG_DOUBLE_SPARSE)
            : (use_dict ? (dense ? YT_VSEG_INT_DICT_DENSE : YT_VSEG_INT_DICT_SPARSE)
                        : (dense ? YT_VSEG_INT_DIRECT_DENSE : YT_VSEG_INT_DIRECT_SPARSE));
        V.flags = value_agg ? YT_VSEG_F_AGGREGATE : 0;
        V.data_size = vb.size();
        void* vp = malloc(vb.size() ? vb.size() : 1);
        memcpy(vp, vb.bytes.data(), vb.size());
        V.data = vp;
    }

    out->ts_seg_count = nseg;
    out->val_seg_count = nseg;
    out->ts_segs = tsegs;
    out->val_segs = vsegs;
    return YT_OK;
}

extern "C" int yt_encode_versioned_int64(
    const uint32_t* writes_per_row, const uint64_t* write_ts,
    const int64_t* values, const uint8_t* value_nulls,
    const uint8_t* value_agg,
    const uint32_t* deletes_per_row, const uint64_t* delete_ts,
    int64_t row_count, int64_t max_rows_per_segment,
    YtVersionedColumn* out, char* errbuf, size_t errlen)
{
    return encode_versioned_impl(writes_per_row, write_ts, values, nullptr,
                                 nullptr, nullptr,
                                 value_nulls, value_agg, deletes_per_row,
                                 delete_ts, row_count, max_rows_per_segment,
                                 out, errbuf, errlen);
}

extern "C" int yt_encode_versioned_double(
    const uint32_t* writes_per_row, const uint64_t* write_ts,
    const double* values, const uint8_t* value_nulls,
    const uint8_t* value_agg,
    const uint32_t* deletes_per_row, const uint64_t* delete_ts,
    int64_t row_count, int64_t max_rows_per_segment,
    YtVersionedColumn* out, char* errbuf, size_t errlen)
{
    return encode_versioned_impl(writes_per_row, write_ts, nullptr, values,
                                 nullptr, nullptr,
                                 value_nulls, value_agg, deletes_per_row,
                                 delete_ts, row_count, max_rows_per_segment,
                                 out, errbuf, errlen);
}

/* versioned STRING column (string_column_writer.cpp
 * TVersionedStringColumnWriter; the Any/Composite writers share the byte
 * layout :365-401) */
extern "C" int yt_encode_versioned_string(
    const uint32_t* writes_per_row, const uint64_t* write_ts,
    const char* value_bytes, const uint32_t* value_lens,
    const uint8_t* value_nulls, const uint8_t* value_agg,
    const uint32_t* deletes_per_row, const uint64_t* delete_ts,
    int64_t row_count, int64_t max_rows_per_segment,
    YtVersionedColumn* out, char* errbuf, size_t errlen)
{
    return encode_versioned_impl(writes_per_row, write_ts, nullptr, nullptr,
                                 value_bytes, value_lens,
                                 value_nulls, value_agg, deletes_per_row,
                                 delete_ts, row_count, max_rows_per_segment,
                                 out, errbuf, errlen);
}

extern "C" void yt_versioned_free(YtVersionedColumn* col)
{
    if (!col) return;
    for (int i = 0; i < col->ts_seg_count; i++)
        free((void*)col->ts_segs[i].data);
    for (int i = 0; i < col->val_seg_count; i++)
        free((void*)col->val_segs[i].data);
    free((void*)col->ts_segs);
    free((void*)col->val_segs);
    memset(col, 0, sizeof(*col));
}
