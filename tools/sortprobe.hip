// probe: rocPRIM u32/u32 sort_pairs at n=4.5e8 (end_bit 20) + random 32B gather
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <rocprim/rocprim.hpp>
#define CHK(x) do { hipError_t e=(x); if(e!=hipSuccess){printf("err %s\n", hipGetErrorString(e)); return 1;} } while(0)

__global__ void k_fill(uint32_t* k, uint32_t* v, int64_t n) {
    int64_t gs = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x*blockDim.x+threadIdx.x; i < n; i += gs) {
        uint64_t x = (uint64_t)i * 0x9E3779B97F4A7C15ull;
        x ^= x >> 30; x *= 0xBF58476D1CE4E5B9ull; x ^= x >> 27;
        k[i] = (uint32_t)(x & 0xFFFFF);   // 20-bit keys
        v[i] = (uint32_t)i;
    }
}
struct Rec { uint64_t w[4]; };
__global__ void k_gather(const Rec* rec, const uint32_t* idx, int64_t n, uint64_t* out) {
    uint64_t acc = 0;
    int64_t gs = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x*blockDim.x+threadIdx.x; i < n; i += gs) {
        const Rec& r = rec[idx[i]];
        acc += r.w[0] + r.w[1] + r.w[2] + r.w[3];
    }
    for (int off = 32; off; off >>= 1) acc += __shfl_down((unsigned long long)acc, off, 64);
    if ((threadIdx.x & 63) == 0) atomicAdd((unsigned long long*)out, (unsigned long long)acc);
}
int main() {
    const int64_t N = 450000000;
    uint32_t *k, *v, *k2, *v2; uint64_t* out;
    CHK(hipMalloc(&k, N*4)); CHK(hipMalloc(&v, N*4));
    CHK(hipMalloc(&k2, N*4)); CHK(hipMalloc(&v2, N*4)); CHK(hipMalloc(&out, 8));
    Rec* rec; CHK(hipMalloc(&rec, N*32));
    hipLaunchKernelGGL(k_fill, dim3(4096), dim3(256), 0, 0, k, v, N);
    CHK(hipMemset(rec, 1, (size_t)N*32));
    CHK(hipDeviceSynchronize());
    size_t tb = 0;
    CHK(rocprim::radix_sort_pairs(nullptr, tb, k, k2, v, v2, (size_t)N, 0, 20));
    void* tmp; CHK(hipMalloc(&tmp, tb));
    hipEvent_t e0, e1; CHK(hipEventCreate(&e0)); CHK(hipEventCreate(&e1));
    for (int rep = 0; rep < 3; rep++) {
        CHK(hipEventRecord(e0));
        CHK(rocprim::radix_sort_pairs(tmp, tb, k, k2, v, v2, (size_t)N, 0, 20));
        CHK(hipEventRecord(e1)); CHK(hipEventSynchronize(e1));
        float ms; CHK(hipEventElapsedTime(&ms, e0, e1));
        printf("sort20 u32/u32 n=4.5e8: %.2f ms (tmp %.1f GB)\n", ms, tb/1e9);
    }
    for (int rep = 0; rep < 3; rep++) {
        CHK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_gather, dim3(8192), dim3(256), 0, 0, rec, v2, N, out);
        CHK(hipEventRecord(e1)); CHK(hipEventSynchronize(e1));
        float ms; CHK(hipEventElapsedTime(&ms, e0, e1));
        printf("gather32B n=4.5e8 sorted-idx: %.2f ms\n", ms);
    }
    return 0;
}
