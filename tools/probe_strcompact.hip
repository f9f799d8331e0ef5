/* probe: why does k_strgrp_compact-style slot-table compaction run at
 * ~80 GB/s? Variants over table size, grid size, struct vs rep-only read,
 * and with/without the key-byte pool copy. hipcc --offload-arch=gfx950. */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <random>

struct StrSlot { unsigned long long rep; unsigned long long sum_bits;
                 unsigned cnt; unsigned nonnull; };
struct OutG { unsigned long long pool_off; unsigned len, pad_;
              unsigned long long sum_bits; unsigned long long cnt, nonnull; };

#define CHK(x) do { hipError_t e=(x); if(e){printf("HIP %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while(0)

__global__ void k_fill(StrSlot* slots, const unsigned long long* idx, long long m)
{
    for (long long i = (long long)blockIdx.x*blockDim.x+threadIdx.x; i < m;
         i += (long long)gridDim.x*blockDim.x) {
        StrSlot& s = slots[idx[i]];
        s.rep = (1ULL<<32) | (unsigned long long)(i + 1);
        s.sum_bits = i; s.cnt = 2; s.nonnull = 2;
    }
}

/* V0: mirror of the shipped kernel (ballot + scan + wave-aggregated atomics,
 * 10-byte payload copy from a fake dict blob) */
__global__ void k_v0(const StrSlot* slots, unsigned long long n,
                     const char* dict, unsigned long long dictsz,
                     OutG* out, unsigned long long* ctr,
                     char* pool, unsigned long long* pcur, int do_copy)
{
    const int lane = threadIdx.x & 63;
    for (unsigned long long i = (unsigned long long)blockIdx.x*blockDim.x+threadIdx.x;
         i < n; i += (unsigned long long)gridDim.x*blockDim.x) {
        const StrSlot& sl = slots[i];
        const bool occ = sl.rep != 0;
        unsigned len = 0; const char* p = nullptr;
        if (occ) {
            unsigned long long id = sl.rep & 0xFFFFFFFFULL;
            p = dict + (id * 11410905013ULL) % (dictsz - 16);
            len = 10;
        }
        unsigned long long mask = __ballot(occ);
        if (!mask) continue;
        unsigned long long run = len;
        #pragma unroll
        for (int d = 1; d < 64; d <<= 1) {
            unsigned long long v = __shfl_up(run, d, 64);
            if (lane >= d) run += v;
        }
        unsigned long long excl = run - len;
        unsigned long long total = __shfl(run, 63, 64);
        int leader = __ffsll(mask) - 1;
        unsigned long long pb = 0, cb = 0;
        if (lane == leader) {
            pb = atomicAdd(pcur, total);
            cb = atomicAdd(ctr, (unsigned long long)__popcll(mask));
        }
        pb = __shfl(pb, leader, 64);
        cb = __shfl(cb, leader, 64);
        if (!occ) continue;
        unsigned long long off = pb + excl;
        if (do_copy) for (unsigned k = 0; k < len; k++) pool[off+k] = p[k];
        unsigned long long idx = cb + __popcll(mask & ((1ULL<<lane)-1));
        OutG& g = out[idx];
        g.pool_off = off; g.len = len; g.pad_ = 0;
        g.sum_bits = sl.sum_bits; g.cnt = sl.cnt; g.nonnull = sl.nonnull;
    }
}

/* V1: read rep via __ldg-style scalarized first pass (same kernel, but rely
 * on reading only .rep for empties by loading it separately) */
__global__ void k_v1(const StrSlot* slots, unsigned long long n,
                     OutG* out, unsigned long long* ctr)
{
    for (unsigned long long i = (unsigned long long)blockIdx.x*blockDim.x+threadIdx.x;
         i < n; i += (unsigned long long)gridDim.x*blockDim.x) {
        unsigned long long rep = slots[i].rep;
        if (rep == 0) continue;
        unsigned long long idx = atomicAdd(ctr, 1ULL);
        OutG& g = out[idx];
        g.pool_off = 0; g.len = 10; g.pad_ = 0;
        g.sum_bits = slots[i].sum_bits; g.cnt = slots[i].cnt;
        g.nonnull = slots[i].nonnull;
    }
}

/* V2: pure rep scan, count only (upper bound on scan speed) */
__global__ void k_v2(const StrSlot* slots, unsigned long long n,
                     unsigned long long* ctr)
{
    unsigned long long c = 0;
    for (unsigned long long i = (unsigned long long)blockIdx.x*blockDim.x+threadIdx.x;
         i < n; i += (unsigned long long)gridDim.x*blockDim.x)
        c += (slots[i].rep != 0);
    if (c) atomicAdd(ctr, c);
}

static float run1(void (*launch)(int, hipStream_t), int grid)
{
    hipEvent_t a, b; CHK(hipEventCreate(&a)); CHK(hipEventCreate(&b));
    CHK(hipEventRecord(a, 0));
    launch(grid, 0);
    CHK(hipEventRecord(b, 0));
    CHK(hipDeviceSynchronize());
    float ms; CHK(hipEventElapsedTime(&ms, a, b));
    hipEventDestroy(a); hipEventDestroy(b);
    return ms;
}

StrSlot* d_slots; OutG* d_out; unsigned long long* d_ctr; char* d_pool; char* d_dict;
unsigned long long N, DICTSZ = 1ULL<<30;

int main(int argc, char** argv)
{
    N = argc > 1 ? strtoull(argv[1], 0, 10) : (1ULL<<28);
    long long M = argc > 2 ? atoll(argv[2]) : 6300000;
    printf("N=%llu slots (%.2f GB), M=%lld occupied\n", N, N*24.0/1e9, M);
    CHK(hipMalloc(&d_slots, N*sizeof(StrSlot)));
    CHK(hipMemset(d_slots, 0, N*sizeof(StrSlot)));
    CHK(hipMalloc(&d_out, (M+64)*sizeof(OutG)));
    CHK(hipMalloc(&d_ctr, 16));
    CHK(hipMalloc(&d_pool, M*16+1024));
    CHK(hipMalloc(&d_dict, DICTSZ));
    /* scatter M occupied slots */
    std::vector<unsigned long long> idx(M);
    std::mt19937_64 rng(42);
    for (long long i = 0; i < M; i++) idx[i] = rng() % N;
    unsigned long long* d_idx; CHK(hipMalloc(&d_idx, M*8));
    CHK(hipMemcpy(d_idx, idx.data(), M*8, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_fill, dim3(2048), dim3(256), 0, 0, d_slots, d_idx, M);
    CHK(hipDeviceSynchronize());

    for (int rep = 0; rep < 2; rep++) {
        for (int grid : {2048, 8192, 32768}) {
            CHK(hipMemset(d_ctr, 0, 16));
            float v0 = run1([](int g, hipStream_t s){
                hipLaunchKernelGGL(k_v0, dim3(g), dim3(256), 0, s, d_slots, N,
                                   d_dict, DICTSZ, d_out, d_ctr, d_pool, d_ctr+1, 1);}, grid);
            CHK(hipMemset(d_ctr, 0, 16));
            float v0nc = run1([](int g, hipStream_t s){
                hipLaunchKernelGGL(k_v0, dim3(g), dim3(256), 0, s, d_slots, N,
                                   d_dict, DICTSZ, d_out, d_ctr, d_pool, d_ctr+1, 0);}, grid);
            CHK(hipMemset(d_ctr, 0, 16));
            float v1 = run1([](int g, hipStream_t s){
                hipLaunchKernelGGL(k_v1, dim3(g), dim3(256), 0, s, d_slots, N, d_out, d_ctr);}, grid);
            CHK(hipMemset(d_ctr, 0, 16));
            float v2 = run1([](int g, hipStream_t s){
                hipLaunchKernelGGL(k_v2, dim3(g), dim3(256), 0, s, d_slots, N, d_ctr);}, grid);
            unsigned long long c; CHK(hipMemcpy(&c, d_ctr, 8, hipMemcpyDeviceToHost));
            double gb = N*24.0/1e9;
            printf("grid=%5d  v0=%7.2fms (%6.1f GB/s)  v0_nocopy=%7.2fms  "
                   "v1=%7.2fms  v2_scan=%7.2fms (%6.1f GB/s)  cnt=%llu\n",
                   grid, v0, gb/(v0/1e3), v0nc, v1, v2, gb/(v2/1e3), c);
        }
    }
    return 0;
}
