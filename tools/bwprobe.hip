// tools/bwprobe.hip — streaming-bandwidth ceiling probe for the histo/scatter
// launch shapes (NOT part of the product .so). Measures GB/s for:
//   k_sum1    : sum one int64 col            (pure read)
//   k_sum4    : sum four int64 cols          (multi-stream read)
//   k_pred    : 3-predicate + bucketid write (histo-shaped, no LDS)
// each at R rows-per-lane (1/2/4, vectorized where R>=2), T in {256,512},
// grid in {2048, 4096, 8192}.
//   hipcc --offload-arch=gfx950 -O3 tools/bwprobe.hip -o gpurun_out/bwprobe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <cstdlib>

#define CHK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    fprintf(stderr, "%s:%d %s\n", __FILE__, __LINE__, hipGetErrorString(e)); \
    exit(1); } } while (0)

typedef long long ll2 __attribute__((ext_vector_type(2)));

__global__ void k_fill(int64_t* p, int64_t n, uint64_t seed) {
    int64_t gs = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += gs) {
        uint64_t x = seed + (uint64_t)i * 0x9E3779B97F4A7C15ull;
        x ^= x >> 30; x *= 0xBF58476D1CE4E5B9ull; x ^= x >> 27;
        p[i] = (int64_t)(x & 0x7FFFFFFFull);
    }
}

template <int R>
__global__ void k_sum1(const int64_t* a, int64_t n, int64_t* out) {
    int64_t acc = 0;
    int64_t stride = (int64_t)gridDim.x * blockDim.x * R;
    for (int64_t base = ((int64_t)blockIdx.x * blockDim.x) * R; base < n;
         base += stride) {
        int64_t i = base + threadIdx.x;
        #pragma unroll
        for (int k = 0; k < R; k++) {
            int64_t j = i + (int64_t)k * blockDim.x;
            if (j < n) acc += a[j];
        }
    }
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if ((threadIdx.x & 63) == 0) atomicAdd((unsigned long long*)out, (unsigned long long)acc);
}

template <int R>
__global__ void k_sum4(const int64_t* a, const int64_t* b, const int64_t* c,
                       const int64_t* d, int64_t n, int64_t* out) {
    int64_t acc = 0;
    int64_t stride = (int64_t)gridDim.x * blockDim.x * R;
    for (int64_t base = ((int64_t)blockIdx.x * blockDim.x) * R; base < n;
         base += stride) {
        int64_t i = base + threadIdx.x;
        #pragma unroll
        for (int k = 0; k < R; k++) {
            int64_t j = i + (int64_t)k * blockDim.x;
            if (j < n) acc += a[j] + b[j] + c[j] + d[j];
        }
    }
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if ((threadIdx.x & 63) == 0) atomicAdd((unsigned long long*)out, (unsigned long long)acc);
}

/* histo-shaped: pred over 3 cols, write a bucketid (no LDS histogram) */
template <int R>
__global__ void k_pred(const int64_t* a, const int64_t* b, const int64_t* c,
                       int64_t n, uint16_t* bid, int64_t* out) {
    int64_t acc = 0;
    int64_t stride = (int64_t)gridDim.x * blockDim.x * R;
    for (int64_t base = ((int64_t)blockIdx.x * blockDim.x) * R; base < n;
         base += stride) {
        int64_t i = base + threadIdx.x;
        #pragma unroll
        for (int k = 0; k < R; k++) {
            int64_t j = i + (int64_t)k * blockDim.x;
            if (j >= n) continue;
            int64_t va = a[j], vb = b[j], vc = c[j];
            bool pass = va < (1ll << 30) && vb < (int64_t)((1u << 31) * 0.9) &&
                        (vc & 63) != 63;
            uint64_t h = (uint64_t)vc * 0x9E3779B97F4A7C15ull;
            bid[j] = pass ? (uint16_t)(h >> 52) : (uint16_t)0xFFFF;
            acc += pass;
        }
    }
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if ((threadIdx.x & 63) == 0) atomicAdd((unsigned long long*)out, (unsigned long long)acc);
}

static double bench(void (*launch)(int, int), int grid, int threads, double gb) {
    hipEvent_t e0, e1;
    CHK(hipEventCreate(&e0)); CHK(hipEventCreate(&e1));
    launch(grid, threads);  // warmup
    CHK(hipDeviceSynchronize());
    float best = 1e30f;
    for (int rep = 0; rep < 3; rep++) {
        CHK(hipEventRecord(e0));
        launch(grid, threads);
        CHK(hipEventRecord(e1));
        CHK(hipEventSynchronize(e1));
        float ms; CHK(hipEventElapsedTime(&ms, e0, e1));
        if (ms < best) best = ms;
    }
    CHK(hipEventDestroy(e0)); CHK(hipEventDestroy(e1));
    return gb / (best / 1e3);
}

int main() {
    const int64_t N = 1000LL * 1000 * 1000;
    int64_t *a, *b, *c, *d, *out;
    uint16_t* bid;
    CHK(hipMalloc(&a, N * 8)); CHK(hipMalloc(&b, N * 8));
    CHK(hipMalloc(&c, N * 8)); CHK(hipMalloc(&d, N * 8));
    CHK(hipMalloc(&bid, N * 2)); CHK(hipMalloc(&out, 8));
    hipLaunchKernelGGL(k_fill, dim3(4096), dim3(256), 0, 0, a, N, 1);
    hipLaunchKernelGGL(k_fill, dim3(4096), dim3(256), 0, 0, b, N, 2);
    hipLaunchKernelGGL(k_fill, dim3(4096), dim3(256), 0, 0, c, N, 3);
    hipLaunchKernelGGL(k_fill, dim3(4096), dim3(256), 0, 0, d, N, 4);
    CHK(hipDeviceSynchronize());

    static int64_t *A, *B, *C, *D, *OUT; static uint16_t* BID; static int64_t NN;
    A = a; B = b; C = c; D = d; OUT = out; BID = bid; NN = N;

    struct Case { const char* name; void (*fn)(int, int); double gb; };
    #define L(kern, R, ...) +[](int g, int t) { \
        hipLaunchKernelGGL(kern<R>, dim3(g), dim3(t), 0, 0, __VA_ARGS__); }
    Case cases[] = {
        {"sum1 R1", L(k_sum1, 1, A, NN, OUT), 8.0},
        {"sum1 R2", L(k_sum1, 2, A, NN, OUT), 8.0},
        {"sum1 R4", L(k_sum1, 4, A, NN, OUT), 8.0},
        {"sum4 R1", L(k_sum4, 1, A, B, C, D, NN, OUT), 32.0},
        {"sum4 R2", L(k_sum4, 2, A, B, C, D, NN, OUT), 32.0},
        {"sum4 R4", L(k_sum4, 4, A, B, C, D, NN, OUT), 32.0},
        {"pred R1", L(k_pred, 1, A, B, C, NN, BID, OUT), 26.0},
        {"pred R2", L(k_pred, 2, A, B, C, NN, BID, OUT), 26.0},
        {"pred R4", L(k_pred, 4, A, B, C, NN, BID, OUT), 26.0},
    };
    for (auto& cs : cases) {
        for (int t : {256, 512}) {
            for (int g : {2048, 4096, 8192}) {
                double gbs = bench(cs.fn, g, t, cs.gb);
                printf("%-8s T=%-4d G=%-5d %8.0f GB/s\n", cs.name, t, g, gbs);
            }
        }
        fflush(stdout);
    }
    return 0;
}
