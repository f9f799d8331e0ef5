// Byte-fixture generator: compiles the REFERENCE's own BitPackUnsignedVector
// (yt/yt/core/misc/bit_packed_unsigned_vector-inl.h, the format-defining
// bit-pack codec) from the sources where they lie under /root/reference and
// prints packed bytes for deterministic vectors. Built by oracle/Makefile
// into oracle/_ref/ (gitignored; the JSON fixture it generates is committed
// as tests/golden/bitpack_ref.json). /root/reference exists only in the
// build container — the GPU box never runs this.
#include <yt/yt/core/misc/bit_packed_unsigned_vector.h>
#include <cstdio>
#include <vector>

using namespace NYT;

static unsigned long long xs(unsigned long long& s)
{
    s ^= s << 13; s ^= s >> 7; s ^= s << 17;
    return s;
}

int main()
{
    printf("[\n");
    bool first = true;
    // sweep widths 0..64 plus edge shapes
    for (int width = 0; width <= 64; width++) {
        for (int n : {1, 7, 64, 129}) {
            unsigned long long seed = 0x9E3779B97F4A7C15ULL ^ (width * 1315423911u + n);
            std::vector<ui64> v(n);
            ui64 maxv = width == 0 ? 0 : (width == 64 ? ~0ULL : ((1ULL << width) - 1));
            for (auto& x : v) x = width == 0 ? 0 : (xs(seed) & maxv);
            if (n > 0 && width > 0) v[0] = maxv;   /* pin the top value */
            auto ref = BitPackUnsignedVector(TRange<ui64>(v.data(), v.size()), maxv);
            if (!first) printf(",\n");
            first = false;
            printf("{\"width\":%d,\"n\":%d,\"maxv\":%llu,\"values\":[", width, n,
                   (unsigned long long)maxv);
            for (int i = 0; i < n; i++)
                printf("%s%llu", i ? "," : "", (unsigned long long)v[i]);
            printf("],\"bytes\":\"");
            for (size_t i = 0; i < ref.Size(); i++)
                printf("%02x", (unsigned char)ref[i]);
            printf("\"}");
        }
    }
    printf("\n]\n");
    return 0;
}
