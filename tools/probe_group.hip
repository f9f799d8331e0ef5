/* probe_group.hip — standalone microbenchmark isolating the cost layers of
 * the fused scan→group kernel on gfx950:
 *   V0 decode-only          (bit-unpack from LDS tile, sum into registers)
 *   V1 + table probe        (relaxed load of the key slot, no update)
 *   V2 + CAS probe          (atomicCAS claim like the product kernel)
 *   V3 probe + 1 atomicAdd
 *   V4 probe + 3 atomicAdd  (the product's current update)
 *   V5 relaxed-probe + 2 atomicAdd (candidate design)
 *
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/probe_group.hip -o tools/probe_group
 * Run:   ./tools/probe_group [rows_millions]
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <cstdlib>

#define CHECK(x) do { hipError_t e = (x); if (e) { printf("hip err %s @%d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

constexpr int TILE = 4096;
constexpr uint32_t WK = 21;   // key width (1M keys zigzag)
constexpr uint32_t WV = 41;   // value width

__device__ __forceinline__ uint64_t mix64(uint64_t x)
{
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

__device__ __forceinline__ uint64_t bp_get_win(const uint64_t* win, uint32_t width,
                                               uint64_t index, uint64_t w0)
{
    uint64_t bit = index * width;
    uint64_t wi = (bit >> 6) - w0;
    unsigned off = (unsigned)(bit & 63);
    uint64_t w1 = win[wi] >> off;
    if (off + width > 64) {
        w1 |= (win[wi + 1] & ((1ULL << ((off + width) & 63)) - 1)) << (64 - off);
    } else {
        w1 &= (1ULL << width) - 1;
    }
    return w1;
}

/* fill packed arrays with pseudo-random bits (keys in [1, 2^20]) */
__global__ void k_fill(uint64_t* kw, uint64_t* vw, int64_t nkw, int64_t nvw)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < nkw) kw[i] = mix64(i * 2654435761ULL);
    if (i < nvw) vw[i] = mix64(i ^ 0xABCDEF0123456789ULL);
}

template <int VARIANT>
__global__ void __launch_bounds__(256)
k_scan(const uint64_t* kwords, const uint64_t* vwords, int64_t rows,
       unsigned long long* slots, uint64_t mask, int stride,
       unsigned long long* out)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    const int64_t ntiles = (rows + TILE - 1) / TILE;
    uint64_t acc = 0;

    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t t0 = tile * TILE;
        int64_t t1 = t0 + TILE < rows ? t0 + TILE : rows;

        int64_t kW0 = (uint64_t)t0 * WK >> 6;
        int64_t kW1 = (((uint64_t)t1 * WK) + 63) >> 6;
        uint64_t* klds = (uint64_t*)smem;
        int64_t knw = kW1 - kW0 + 1;
        for (int64_t i = tid; i < knw; i += 256) klds[i] = kwords[kW0 + i];

        int64_t vW0 = (uint64_t)t0 * WV >> 6;
        int64_t vW1 = (((uint64_t)t1 * WV) + 63) >> 6;
        uint64_t* vlds = klds + knw + 2;
        int64_t vnw = vW1 - vW0 + 1;
        for (int64_t i = tid; i < vnw; i += 256) vlds[i] = vwords[vW0 + i];
        __syncthreads();

        const int R = TILE / 256;
        for (int i = 0; i < R; i++) {
            int64_t j = t0 + (int64_t)i * 256 + tid;
            if (j >= t1) break;
            uint64_t key = bp_get_win(klds, WK, j, kW0) & ((1u << 20) - 1);
            uint64_t val = bp_get_win(vlds, WV, j, vW0);
            key += 1;   // avoid 0

            if (VARIANT == 0) {
                acc += key + val;
            } else {
                uint64_t h = mix64(key);
                uint64_t s = h & mask;
                unsigned long long* slot = nullptr;
                for (int it = 0; it < 64; it++) {
                    unsigned long long k;
                    if (VARIANT == 2 || VARIANT == 3 || VARIANT == 4) {
                        k = atomicCAS(slots + s * stride, 0ULL, (unsigned long long)key);
                        if (k == 0ULL || k == (unsigned long long)key) { slot = slots + s * stride; break; }
                    } else {
                        k = __hip_atomic_load(slots + s * stride, __ATOMIC_RELAXED,
                                              __HIP_MEMORY_SCOPE_AGENT);
                        if (k == (unsigned long long)key) { slot = slots + s * stride; break; }
                        if (k == 0ULL) {
                            k = atomicCAS(slots + s * stride, 0ULL, (unsigned long long)key);
                            if (k == 0ULL || k == (unsigned long long)key) { slot = slots + s * stride; break; }
                        }
                    }
                    s = (s + 1) & mask;
                }
                if (!slot) continue;
                if (VARIANT == 1) { acc += (uint64_t)(slot - slots); }
                if (VARIANT == 3) { atomicAdd(slot + 1, val); }
                if (VARIANT == 4) {
                    atomicAdd(slot + 1, 1ULL);
                    atomicAdd(slot + 2, val);
                    atomicAdd(slot + 3, 1ULL);
                }
                if (VARIANT == 5) {
                    atomicAdd(slot + 1, 1ULL);
                    atomicAdd(slot + 2, val);
                }
            }
        }
        __syncthreads();
    }
    if (acc) atomicAdd(out, acc);   // keep V0/V1 alive
}

/* V8: decode + aggregate into a per-WG LDS table (2048 slots, keys masked
 * to 11 bits), LDS atomics, flush to global at end. Measures the LDS-table
 * aggregation ceiling. */
__global__ void __launch_bounds__(256)
k_scan_lds(const uint64_t* kwords, const uint64_t* vwords, int64_t rows,
           unsigned long long* slots, unsigned long long* out)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    const int64_t ntiles = (rows + TILE - 1) / TILE;
    constexpr int LSLOTS = 2048;
    /* LDS layout: [ktile words][vtile words][table: key,cnt,sum x LSLOTS] */
    uint64_t* klds = (uint64_t*)smem;

    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t t0 = tile * TILE;
        int64_t t1 = t0 + TILE < rows ? t0 + TILE : rows;
        int64_t kW0 = (uint64_t)t0 * WK >> 6;
        int64_t kW1 = (((uint64_t)t1 * WK) + 63) >> 6;
        int64_t knw = kW1 - kW0 + 1;
        for (int64_t i = tid; i < knw; i += 256) klds[i] = kwords[kW0 + i];
        int64_t vW0 = (uint64_t)t0 * WV >> 6;
        int64_t vW1 = (((uint64_t)t1 * WV) + 63) >> 6;
        uint64_t* vlds = klds + knw + 2;
        int64_t vnw = vW1 - vW0 + 1;
        for (int64_t i = tid; i < vnw; i += 256) vlds[i] = vwords[vW0 + i];
        unsigned long long* tab = (unsigned long long*)(vlds + vnw + 2);
        for (int i = tid; i < LSLOTS * 2; i += 256) tab[i] = 0;
        __syncthreads();

        const int R = TILE / 256;
        for (int i = 0; i < R; i++) {
            int64_t j = t0 + (int64_t)i * 256 + tid;
            if (j >= t1) break;
            uint64_t key = bp_get_win(klds, WK, j, kW0) & (LSLOTS - 1);
            uint64_t val = bp_get_win(vlds, WV, j, vW0);
            /* direct-indexed LDS slot (key==slot here); 2 LDS atomics */
            atomicAdd(&tab[key * 2], 1ULL);
            atomicAdd(&tab[key * 2 + 1], val);
        }
        __syncthreads();
        /* flush: 2 global atomics per occupied slot per WG-tile */
        for (int i = tid; i < LSLOTS; i += 256) {
            unsigned long long c = tab[i * 2];
            if (c) {
                atomicAdd(slots + (uint64_t)i * 6 + 1, c);
                atomicAdd(slots + (uint64_t)i * 6 + 2, tab[i * 2 + 1]);
            }
        }
        __syncthreads();
    }
    (void)out;
}

/* V9: decode + write (key,val) 16B/row to a per-tile contiguous region —
 * the partition-scatter lower bound (no bucketing). */
__global__ void __launch_bounds__(256)
k_scan_scatter(const uint64_t* kwords, const uint64_t* vwords, int64_t rows,
               ulonglong2* outbuf)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    const int64_t ntiles = (rows + TILE - 1) / TILE;
    uint64_t* klds = (uint64_t*)smem;
    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t t0 = tile * TILE;
        int64_t t1 = t0 + TILE < rows ? t0 + TILE : rows;
        int64_t kW0 = (uint64_t)t0 * WK >> 6;
        int64_t kW1 = (((uint64_t)t1 * WK) + 63) >> 6;
        int64_t knw = kW1 - kW0 + 1;
        for (int64_t i = tid; i < knw; i += 256) klds[i] = kwords[kW0 + i];
        int64_t vW0 = (uint64_t)t0 * WV >> 6;
        int64_t vW1 = (((uint64_t)t1 * WV) + 63) >> 6;
        uint64_t* vlds = klds + knw + 2;
        int64_t vnw = vW1 - vW0 + 1;
        for (int64_t i = tid; i < vnw; i += 256) vlds[i] = vwords[vW0 + i];
        __syncthreads();
        const int R = TILE / 256;
        for (int i = 0; i < R; i++) {
            int64_t j = t0 + (int64_t)i * 256 + tid;
            if (j >= t1) break;
            uint64_t key = bp_get_win(klds, WK, j, kW0);
            uint64_t val = bp_get_win(vlds, WV, j, vW0);
            outbuf[j] = make_ulonglong2(key, val);
        }
        __syncthreads();
    }
}

template <int VARIANT>
float run(const uint64_t* kw, const uint64_t* vw, int64_t rows,
          unsigned long long* slots, uint64_t nslots, int stride,
          unsigned long long* out, int iters)
{
    int64_t ntiles = (rows + TILE - 1) / TILE;
    int grid = ntiles < 2048 ? (int)ntiles : 2048;
    size_t lds = ((size_t)TILE * WK / 64 + 2 + (size_t)TILE * WV / 64 + 4) * 8 + 64;
    hipEvent_t e0, e1;
    CHECK(hipEventCreate(&e0));
    CHECK(hipEventCreate(&e1));
    // warmup
    CHECK(hipMemset(slots, 0, nslots * stride * 8));
    hipLaunchKernelGGL(k_scan<VARIANT>, dim3(grid), dim3(256), lds, 0,
                       kw, vw, rows, slots, nslots - 1, stride, out);
    CHECK(hipDeviceSynchronize());
    float total = 0;
    for (int i = 0; i < iters; i++) {
        CHECK(hipMemset(slots, 0, nslots * stride * 8));
        CHECK(hipDeviceSynchronize());
        CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_scan<VARIANT>, dim3(grid), dim3(256), lds, 0,
                           kw, vw, rows, slots, nslots - 1, stride, out);
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        float ms;
        CHECK(hipEventElapsedTime(&ms, e0, e1));
        total += ms;
    }
    CHECK(hipEventDestroy(e0));
    CHECK(hipEventDestroy(e1));
    return total / iters;
}


/* ================= probe round 2: partition design ================= */

/* V8b: PERSISTENT direct-indexed LDS table (2048 slots) across all of a
 * block's tiles; flush once at block end → pure LDS-atomic ceiling. */
__global__ void __launch_bounds__(256)
k_scan_lds_persist(const uint64_t* kwords, const uint64_t* vwords, int64_t rows,
                   unsigned long long* slots)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    const int64_t ntiles = (rows + TILE - 1) / TILE;
    constexpr int LSLOTS = 2048;
    /* layout: [table 2*LSLOTS u64][tile staging] */
    unsigned long long* tab = (unsigned long long*)smem;
    uint64_t* stage = (uint64_t*)smem + 2 * LSLOTS;
    for (int i = tid; i < LSLOTS * 2; i += 256) tab[i] = 0;
    __syncthreads();

    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t t0 = tile * TILE;
        int64_t t1 = t0 + TILE < rows ? t0 + TILE : rows;
        int64_t kW0 = (uint64_t)t0 * WK >> 6;
        int64_t kW1 = (((uint64_t)t1 * WK) + 63) >> 6;
        int64_t knw = kW1 - kW0 + 1;
        uint64_t* klds = stage;
        for (int64_t i = tid; i < knw; i += 256) klds[i] = kwords[kW0 + i];
        int64_t vW0 = (uint64_t)t0 * WV >> 6;
        int64_t vW1 = (((uint64_t)t1 * WV) + 63) >> 6;
        uint64_t* vlds = klds + knw + 2;
        int64_t vnw = vW1 - vW0 + 1;
        for (int64_t i = tid; i < vnw; i += 256) vlds[i] = vwords[vW0 + i];
        __syncthreads();
        const int R = TILE / 256;
        for (int i = 0; i < R; i++) {
            int64_t j = t0 + (int64_t)i * 256 + tid;
            if (j >= t1) break;
            uint64_t key = bp_get_win(klds, WK, j, kW0) & (LSLOTS - 1);
            uint64_t val = bp_get_win(vlds, WV, j, vW0);
            atomicAdd(&tab[key * 2], 1ULL);
            atomicAdd(&tab[key * 2 + 1], val);
        }
        __syncthreads();
    }
    for (int i = tid; i < LSLOTS; i += 256) {
        if (tab[i * 2]) {
            atomicAdd(slots + (uint64_t)i * 6 + 1, tab[i * 2]);
            atomicAdd(slots + (uint64_t)i * 6 + 2, tab[i * 2 + 1]);
        }
    }
}

/* V8c: persistent LDS OPEN-ADDRESSING table (4096 slots, 64-bit keys, CAS
 * claim + 2 adds) — the realistic Phase B inner loop. Keys masked to 2^10
 * distinct to model ~1K keys per bucket. */
__global__ void __launch_bounds__(256)
k_scan_lds_hash(const uint64_t* kwords, const uint64_t* vwords, int64_t rows,
                unsigned long long* slots)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    const int64_t ntiles = (rows + TILE - 1) / TILE;
    constexpr int HSLOTS = 4096;
    unsigned long long* tab = (unsigned long long*)smem;   /* key,cnt,sum */
    uint64_t* stage = (uint64_t*)smem + 3 * HSLOTS;
    for (int i = tid; i < HSLOTS * 3; i += 256) tab[i] = 0;
    __syncthreads();

    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t t0 = tile * TILE;
        int64_t t1 = t0 + TILE < rows ? t0 + TILE : rows;
        int64_t kW0 = (uint64_t)t0 * WK >> 6;
        int64_t kW1 = (((uint64_t)t1 * WK) + 63) >> 6;
        int64_t knw = kW1 - kW0 + 1;
        uint64_t* klds = stage;
        for (int64_t i = tid; i < knw; i += 256) klds[i] = kwords[kW0 + i];
        int64_t vW0 = (uint64_t)t0 * WV >> 6;
        int64_t vW1 = (((uint64_t)t1 * WV) + 63) >> 6;
        uint64_t* vlds = klds + knw + 2;
        int64_t vnw = vW1 - vW0 + 1;
        for (int64_t i = tid; i < vnw; i += 256) vlds[i] = vwords[vW0 + i];
        __syncthreads();
        const int R = TILE / 256;
        for (int i = 0; i < R; i++) {
            int64_t j = t0 + (int64_t)i * 256 + tid;
            if (j >= t1) break;
            uint64_t key = (bp_get_win(klds, WK, j, kW0) & ((1u << 10) - 1)) + 1;
            uint64_t val = bp_get_win(vlds, WV, j, vW0);
            uint64_t s = mix64(key) & (HSLOTS - 1);
            for (;;) {
                unsigned long long k = tab[s * 3];
                if (k == (unsigned long long)key) break;
                if (k == 0ULL) {
                    k = atomicCAS(&tab[s * 3], 0ULL, (unsigned long long)key);
                    if (k == 0ULL || k == (unsigned long long)key) break;
                }
                s = (s + 1) & (HSLOTS - 1);
            }
            atomicAdd(&tab[s * 3 + 1], 1ULL);
            atomicAdd(&tab[s * 3 + 2], val);
        }
        __syncthreads();
    }
    for (int i = tid; i < HSLOTS; i += 256) {
        if (tab[i * 3]) {
            atomicAdd(slots + (uint64_t)(tab[i * 3] & 1023) * 6 + 1, tab[i * 3 + 1]);
            atomicAdd(slots + (uint64_t)(tab[i * 3] & 1023) * 6 + 2, tab[i * 3 + 2]);
        }
    }
}

/* V10: Phase A — decode + 1024-bucket partition scatter with LDS histogram,
 * block-wide prefix, per-row computed destinations. Global per-bucket
 * cursors advanced once per block-tile. */
__global__ void __launch_bounds__(256)
k_partition(const uint64_t* kwords, const uint64_t* vwords, int64_t rows,
            unsigned long long* cursors, ulonglong2* outbuf, int64_t bucket_stride)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    const int64_t ntiles = (rows + TILE - 1) / TILE;
    constexpr int NB = 1024;
    unsigned* hist = (unsigned*)smem;                  /* NB counters */
    unsigned* base = hist + NB;                        /* NB bases */
    uint64_t* stage = (uint64_t*)(base + NB);

    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t t0 = tile * TILE;
        int64_t t1 = t0 + TILE < rows ? t0 + TILE : rows;
        int64_t kW0 = (uint64_t)t0 * WK >> 6;
        int64_t kW1 = (((uint64_t)t1 * WK) + 63) >> 6;
        int64_t knw = kW1 - kW0 + 1;
        uint64_t* klds = stage;
        for (int64_t i = tid; i < knw; i += 256) klds[i] = kwords[kW0 + i];
        int64_t vW0 = (uint64_t)t0 * WV >> 6;
        int64_t vW1 = (((uint64_t)t1 * WV) + 63) >> 6;
        uint64_t* vlds = klds + knw + 2;
        int64_t vnw = vW1 - vW0 + 1;
        for (int64_t i = tid; i < vnw; i += 256) vlds[i] = vwords[vW0 + i];
        for (int i = tid; i < NB; i += 256) hist[i] = 0;
        __syncthreads();

        const int R = TILE / 256;
        unsigned my_off[32];
        unsigned my_b[32];
        for (int i = 0; i < R; i++) {
            int64_t j = t0 + (int64_t)i * 256 + tid;
            if (j >= t1) { my_b[i] = 0xffffffffu; continue; }
            uint64_t key = bp_get_win(klds, WK, j, kW0);
            unsigned b = (unsigned)(mix64(key + 1) & (NB - 1));
            my_b[i] = b;
            my_off[i] = atomicAdd(&hist[b], 1u);
        }
        __syncthreads();
        /* reserve global space per bucket (1 global atomic per bucket/tile) */
        for (int i = tid; i < NB; i += 256) {
            unsigned c = hist[i];
            base[i] = c ? (unsigned)atomicAdd(&cursors[i], (unsigned long long)c) : 0;
        }
        __syncthreads();
        for (int i = 0; i < R; i++) {
            if (my_b[i] == 0xffffffffu) continue;
            int64_t j = t0 + (int64_t)i * 256 + tid;
            uint64_t key = bp_get_win(klds, WK, j, kW0);
            uint64_t val = bp_get_win(vlds, WV, j, vW0);
            int64_t dst = (int64_t)my_b[i] * bucket_stride + base[my_b[i]] + my_off[i];
            outbuf[dst] = make_ulonglong2(key, val);
        }
        __syncthreads();
    }
}

/* V11: Phase B — stream bucket rows (16B coalesced reads) into the
 * persistent LDS hash table. One block per bucket-chunk. */
__global__ void __launch_bounds__(256)
k_bucket_agg(const ulonglong2* inbuf, const unsigned long long* cursors,
             int64_t bucket_stride, unsigned long long* slots)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    constexpr int HSLOTS = 4096;
    unsigned long long* tab = (unsigned long long*)smem;
    for (int i = tid; i < HSLOTS * 3; i += 256) tab[i] = 0;
    __syncthreads();

    int bucket = blockIdx.x;      /* 1024 buckets, 1024 blocks */
    int64_t n = (int64_t)cursors[bucket];
    const ulonglong2* rows = inbuf + (int64_t)bucket * bucket_stride;
    for (int64_t i = tid; i < n; i += 256) {
        ulonglong2 kv = rows[i];
        uint64_t s = mix64(kv.x) & (HSLOTS - 1);
        for (;;) {
            unsigned long long k = tab[s * 3];
            if (k == (unsigned long long)kv.x) break;
            if (k == 0ULL) {
                k = atomicCAS(&tab[s * 3], 0ULL, (unsigned long long)kv.x);
                if (k == 0ULL || k == (unsigned long long)kv.x) break;
            }
            s = (s + 1) & (HSLOTS - 1);
        }
        atomicAdd(&tab[s * 3 + 1], 1ULL);
        atomicAdd(&tab[s * 3 + 2], kv.y);
    }
    __syncthreads();
    for (int i = tid; i < HSLOTS; i += 256) {
        if (tab[i * 3]) {
            uint64_t g = mix64(tab[i * 3]) & ((1ull << 21) - 1);
            atomicAdd(slots + g * 6 + 1, tab[i * 3 + 1]);
            atomicAdd(slots + g * 6 + 2, tab[i * 3 + 2]);
        }
    }
}

/* V12: V0-style decode-only but with RUNTIME widths + branchless funnel
 * extraction (mask hoisted per tile) — isolates the cost of runtime vs
 * compile-time widths. */
__device__ __forceinline__ uint64_t bp_get_bl(const uint64_t* win, uint64_t mask,
                                              uint32_t width, uint64_t index, uint64_t w0)
{
    uint64_t bit = index * width;
    uint64_t wi = (bit >> 6) - w0;
    unsigned off = (unsigned)(bit & 63);
    uint64_t lo = win[wi] >> off;
    uint64_t hi = off ? (win[wi + 1] << (64 - off)) : 0;
    return (lo | hi) & mask;
}

__global__ void __launch_bounds__(256)
k_scan_rtw(const uint64_t* kwords, const uint64_t* vwords, int64_t rows,
           uint32_t wk, uint32_t wv, int branchless,
           unsigned long long* out)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    const int64_t ntiles = (rows + TILE - 1) / TILE;
    uint64_t acc = 0;
    const uint64_t kmask = (wk >= 64) ? ~0ULL : ((1ULL << wk) - 1);
    const uint64_t vmask = (wv >= 64) ? ~0ULL : ((1ULL << wv) - 1);

    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t t0 = tile * TILE;
        int64_t t1 = t0 + TILE < rows ? t0 + TILE : rows;
        int64_t kW0 = (uint64_t)t0 * wk >> 6;
        int64_t kW1 = (((uint64_t)t1 * wk) + 63) >> 6;
        uint64_t* klds = (uint64_t*)smem;
        int64_t knw = kW1 - kW0 + 1;
        for (int64_t i = tid; i < knw; i += 256) klds[i] = kwords[kW0 + i];
        int64_t vW0 = (uint64_t)t0 * wv >> 6;
        int64_t vW1 = (((uint64_t)t1 * wv) + 63) >> 6;
        uint64_t* vlds = klds + knw + 2;
        int64_t vnw = vW1 - vW0 + 1;
        for (int64_t i = tid; i < vnw; i += 256) vlds[i] = vwords[vW0 + i];
        __syncthreads();
        const int R = TILE / 256;
        for (int i = 0; i < R; i++) {
            int64_t j = t0 + (int64_t)i * 256 + tid;
            if (j >= t1) break;
            uint64_t key, val;
            if (branchless) {
                key = bp_get_bl(klds, kmask, wk, j, kW0);
                val = bp_get_bl(vlds, vmask, wv, j, vW0);
            } else {
                key = bp_get_win(klds, wk, j, kW0);
                val = bp_get_win(vlds, wv, j, vW0);
            }
            acc += key + val;
        }
        __syncthreads();
    }
    if (acc) atomicAdd(out, acc);
}


/* V13: decode straight from GLOBAL (no LDS staging): per-value 1-2 loads,
 * L1 absorbs the window overlap, unrolled rows give MLP. */
__global__ void __launch_bounds__(256)
k_scan_noLDS(const uint64_t* kwords, const uint64_t* vwords, int64_t rows,
             uint32_t wk, uint32_t wv, unsigned long long* out)
{
    uint64_t acc = 0;
    const uint64_t kmask = (wk >= 64) ? ~0ULL : ((1ULL << wk) - 1);
    const uint64_t vmask = (wv >= 64) ? ~0ULL : ((1ULL << wv) - 1);
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (int64_t j = i0; j + 3 * stride < rows; j += 4 * stride) {
        #pragma unroll 4
        for (int u = 0; u < 4; u++) {
            int64_t jj = j + u * stride;
            uint64_t kb = jj * wk;
            uint64_t lo = kwords[kb >> 6] >> (kb & 63);
            uint64_t hi = (kb & 63) ? (kwords[(kb >> 6) + 1] << (64 - (kb & 63))) : 0;
            uint64_t key = (lo | hi) & kmask;
            uint64_t vb = jj * wv;
            uint64_t lo2 = vwords[vb >> 6] >> (vb & 63);
            uint64_t hi2 = (vb & 63) ? (vwords[(vb >> 6) + 1] << (64 - (vb & 63))) : 0;
            uint64_t val = (lo2 | hi2) & vmask;
            acc += key + val;
        }
    }
    if (acc) atomicAdd(out, acc);
}

int main(int argc, char** argv)
{
    int64_t rows = (argc > 1 ? atoll(argv[1]) : 200) * 1000000LL;
    int64_t nkw = rows * WK / 64 + 2, nvw = rows * WV / 64 + 2;
    uint64_t *kw, *vw;
    unsigned long long *slots, *out;
    uint64_t nslots = 1ull << 21;
    int stride = 6;
    CHECK(hipMalloc(&kw, nkw * 8));
    CHECK(hipMalloc(&vw, nvw * 8));
    CHECK(hipMalloc(&slots, nslots * stride * 8));
    CHECK(hipMalloc(&out, 8));
    int64_t nmax = nkw > nvw ? nkw : nvw;
    hipLaunchKernelGGL(k_fill, dim3((nmax + 255) / 256), dim3(256), 0, 0, kw, vw, nkw, nvw);
    CHECK(hipDeviceSynchronize());

    double gb = (rows * (WK + WV) / 8.0) / 1e9;
    const char* names[] = {"decode-only", "relaxed-probe", "cas-probe",
                           "cas+1add", "cas+3add (product)", "relaxed+2add"};
    float ms[6];
    ms[0] = run<0>(kw, vw, rows, slots, nslots, stride, out, 3);
    ms[1] = run<1>(kw, vw, rows, slots, nslots, stride, out, 3);
    ms[2] = run<2>(kw, vw, rows, slots, nslots, stride, out, 3);
    ms[3] = run<3>(kw, vw, rows, slots, nslots, stride, out, 3);
    ms[4] = run<4>(kw, vw, rows, slots, nslots, stride, out, 3);
    ms[5] = run<5>(kw, vw, rows, slots, nslots, stride, out, 3);
    for (int v = 0; v <= 5; v++) {
        printf("V%d %-20s %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               v, names[v], ms[v], gb / ms[v] * 1000, rows / ms[v] / 1e6);
    }

    /* V6/V7: same update shapes against an L2-resident table (2^15 keys,
     * 2^16 slots x 48B = 3 MB) — isolates table-residency effects */
    {
        uint64_t small_nslots = 1ull << 16;
        float m4 = run<4>(kw, vw, rows, slots, small_nslots, stride, out, 3);
        float m5 = run<5>(kw, vw, rows, slots, small_nslots, stride, out, 3);
        printf("V6 cas+3add small-L2    %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               m4, gb / m4 * 1000, rows / m4 / 1e6);
        printf("V7 relaxed+2add smallL2 %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               m5, gb / m5 * 1000, rows / m5 / 1e6);
    }
    /* V8: LDS-table aggregation (2048 direct slots, 2 LDS atomics/row) */
    {
        int64_t ntiles = (rows + TILE - 1) / TILE;
        int grid = ntiles < 2048 ? (int)ntiles : 2048;
        size_t lds = ((size_t)TILE * WK / 64 + 2 + (size_t)TILE * WV / 64 + 4) * 8
                   + 2048 * 2 * 8 + 64;
        hipEvent_t e0, e1;
        CHECK(hipEventCreate(&e0));
        CHECK(hipEventCreate(&e1));
        hipLaunchKernelGGL(k_scan_lds, dim3(grid), dim3(256), lds, 0,
                           kw, vw, rows, slots, out);
        CHECK(hipDeviceSynchronize());
        CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_scan_lds, dim3(grid), dim3(256), lds, 0,
                           kw, vw, rows, slots, out);
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        float ms8;
        CHECK(hipEventElapsedTime(&ms8, e0, e1));
        printf("V8 lds-table 2add       %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               ms8, gb / ms8 * 1000, rows / ms8 / 1e6);
    }
    /* V9: decode + contiguous 16B/row write (partition lower bound) */
    {
        ulonglong2* outbuf;
        CHECK(hipMalloc(&outbuf, rows * 16));
        int64_t ntiles = (rows + TILE - 1) / TILE;
        int grid = ntiles < 2048 ? (int)ntiles : 2048;
        size_t lds = ((size_t)TILE * WK / 64 + 2 + (size_t)TILE * WV / 64 + 4) * 8 + 64;
        hipEvent_t e0, e1;
        CHECK(hipEventCreate(&e0));
        CHECK(hipEventCreate(&e1));
        hipLaunchKernelGGL(k_scan_scatter, dim3(grid), dim3(256), lds, 0,
                           kw, vw, rows, outbuf);
        CHECK(hipDeviceSynchronize());
        CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_scan_scatter, dim3(grid), dim3(256), lds, 0,
                           kw, vw, rows, outbuf);
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        float ms9;
        CHECK(hipEventElapsedTime(&ms9, e0, e1));
        double gb_rw = gb + rows * 16.0 / 1e9;
        printf("V9 decode+16B write     %8.3f ms  %8.1f GB/s(rw %.0f)  %8.2f Grows/s\n",
               ms9, gb / ms9 * 1000, gb_rw / ms9 * 1000, rows / ms9 / 1e6);
        CHECK(hipFree(outbuf));
    }
    /* V8b / V8c: persistent LDS tables */
    {
        int64_t ntiles = (rows + TILE - 1) / TILE;
        int grid = ntiles < 2048 ? (int)ntiles : 2048;
        size_t stage_bytes = ((size_t)TILE * WK / 64 + 2 + (size_t)TILE * WV / 64 + 4) * 8 + 64;
        hipEvent_t e0, e1;
        CHECK(hipEventCreate(&e0));
        CHECK(hipEventCreate(&e1));
        size_t lds_b = 2048 * 2 * 8 + stage_bytes;
        hipLaunchKernelGGL(k_scan_lds_persist, dim3(grid), dim3(256), lds_b, 0, kw, vw, rows, slots);
        CHECK(hipDeviceSynchronize());
        CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_scan_lds_persist, dim3(grid), dim3(256), lds_b, 0, kw, vw, rows, slots);
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        float ms;
        CHECK(hipEventElapsedTime(&ms, e0, e1));
        printf("V8b lds-direct persist  %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               ms, gb / ms * 1000, rows / ms / 1e6);
        size_t lds_c = 4096 * 3 * 8 + stage_bytes;
        hipLaunchKernelGGL(k_scan_lds_hash, dim3(grid), dim3(256), lds_c, 0, kw, vw, rows, slots);
        CHECK(hipDeviceSynchronize());
        CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_scan_lds_hash, dim3(grid), dim3(256), lds_c, 0, kw, vw, rows, slots);
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        CHECK(hipEventElapsedTime(&ms, e0, e1));
        printf("V8c lds-hash persist    %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               ms, gb / ms * 1000, rows / ms / 1e6);
    }
    /* V10 + V11: two-phase partition + bucket aggregate */
    {
        constexpr int NB = 1024;
        int64_t bucket_stride = (rows / NB) * 3 / 2 + 4096;
        ulonglong2* outbuf;
        unsigned long long* cursors;
        CHECK(hipMalloc(&outbuf, (int64_t)NB * bucket_stride * 16));
        CHECK(hipMalloc(&cursors, NB * 8));
        int64_t ntiles = (rows + TILE - 1) / TILE;
        int grid = ntiles < 2048 ? (int)ntiles : 2048;
        size_t stage_bytes = ((size_t)TILE * WK / 64 + 2 + (size_t)TILE * WV / 64 + 4) * 8 + 64;
        size_t lds_a = NB * 4 * 2 + stage_bytes;
        hipEvent_t e0, e1;
        CHECK(hipEventCreate(&e0));
        CHECK(hipEventCreate(&e1));
        CHECK(hipMemset(cursors, 0, NB * 8));
        hipLaunchKernelGGL(k_partition, dim3(grid), dim3(256), lds_a, 0,
                           kw, vw, rows, cursors, outbuf, bucket_stride);
        CHECK(hipDeviceSynchronize());
        CHECK(hipMemset(cursors, 0, NB * 8));
        CHECK(hipDeviceSynchronize());
        CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_partition, dim3(grid), dim3(256), lds_a, 0,
                           kw, vw, rows, cursors, outbuf, bucket_stride);
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        float msA;
        CHECK(hipEventElapsedTime(&msA, e0, e1));
        size_t lds_b2 = 4096 * 3 * 8 + 64;
        CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_bucket_agg, dim3(NB), dim3(256), lds_b2, 0,
                           outbuf, cursors, bucket_stride, slots);
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        float msB;
        CHECK(hipEventElapsedTime(&msB, e0, e1));
        printf("V10 partition (A)       %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               msA, gb / msA * 1000, rows / msA / 1e6);
        printf("V11 bucket agg (B)      %8.3f ms  %8.2f Grows/s\n",
               msB, rows / msB / 1e6);
        printf("V10+V11 total           %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               msA + msB, gb / (msA + msB) * 1000, rows / (msA + msB) / 1e6);
        CHECK(hipFree(outbuf));
        CHECK(hipFree(cursors));
    }
    /* V12: runtime-width decode */
    {
        int64_t ntiles = (rows + TILE - 1) / TILE;
        int grid = ntiles < 2048 ? (int)ntiles : 2048;
        size_t lds = ((size_t)TILE * WK / 64 + 2 + (size_t)TILE * WV / 64 + 4) * 8 + 64;
        hipEvent_t e0, e1;
        CHECK(hipEventCreate(&e0));
        CHECK(hipEventCreate(&e1));
        for (int bl = 0; bl <= 1; bl++) {
            hipLaunchKernelGGL(k_scan_rtw, dim3(grid), dim3(256), lds, 0,
                               kw, vw, rows, WK, WV, bl, out);
            CHECK(hipDeviceSynchronize());
            CHECK(hipEventRecord(e0));
            hipLaunchKernelGGL(k_scan_rtw, dim3(grid), dim3(256), lds, 0,
                               kw, vw, rows, WK, WV, bl, out);
            CHECK(hipEventRecord(e1));
            CHECK(hipEventSynchronize(e1));
            float ms;
            CHECK(hipEventElapsedTime(&ms, e0, e1));
            printf("V12 rtw decode bl=%d    %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
                   bl, ms, gb / ms * 1000, rows / ms / 1e6);
        }
    }
    /* V13 */
    {
        hipEvent_t e0, e1;
        CHECK(hipEventCreate(&e0));
        CHECK(hipEventCreate(&e1));
        hipLaunchKernelGGL(k_scan_noLDS, dim3(2048), dim3(256), 0, 0, kw, vw, rows, WK, WV, out);
        CHECK(hipDeviceSynchronize());
        CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_scan_noLDS, dim3(2048), dim3(256), 0, 0, kw, vw, rows, WK, WV, out);
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        float ms;
        CHECK(hipEventElapsedTime(&ms, e0, e1));
        printf("V13 noLDS rtw decode    %8.3f ms  %8.1f GB/s  %8.2f Grows/s\n",
               ms, gb / ms * 1000, rows / ms / 1e6);
    }
    return 0;
}


