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

__global__ void k_filli32(int32_t* p, int64_t n, uint64_t seed) {
    int64_t gs = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += gs) {
        uint64_t x = seed + (uint64_t)i * 0x9E3779B97F4A7C15ull;
        x ^= x >> 30; x *= 0xBF58476D1CE4E5B9ull; x ^= x >> 27;
        p[i] = (int32_t)(x & 0xFFFF);
    }
}

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

/* ---- incremental reconstruction of k_part_histo ---- */
__device__ __forceinline__ uint64_t mix64(uint64_t x) {
    x ^= x >> 33; x *= 0xFF51AFD7ED558CCDull;
    x ^= x >> 33; x *= 0xC4CEB9FE1A85EC53ull;
    x ^= x >> 33; return x;
}

/* P2: pred + LDS histogram + bucketid (adds the per-survivor LDS atomic) */
template <int R>
__global__ void k_predh(const int64_t* a, const int64_t* b, const int64_t* c,
                        int64_t n, uint32_t P, uint16_t* bid, uint32_t* H,
                        int64_t* out) {
    extern __shared__ uint32_t lhist[];
    for (uint32_t x = threadIdx.x; x < P; x += blockDim.x) lhist[x] = 0;
    __syncthreads();
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
            if (!pass) { bid[j] = 0xFFFFu; continue; }
            uint32_t bk = (uint32_t)(((uint64_t)vc * 0x9E3779B97F4A7C15ull) >> 44)
                          & (P - 1u);
            bid[j] = (uint16_t)bk;
            atomicAdd(&lhist[bk], 1u);
            acc++;
        }
    }
    __syncthreads();
    for (uint32_t x = threadIdx.x; x < P; x += blockDim.x)
        H[(size_t)blockIdx.x * P + x] = lhist[x];
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if ((threadIdx.x & 63) == 0) atomicAdd((unsigned long long*)out, (unsigned long long)acc);
}

/* P3: P2 + real key packing (2 key cols: i64 enc + dict i32, bits-pack, mix hash) */
template <int R>
__global__ void k_predk(const int64_t* a, const int64_t* b, const int64_t* c,
                        const int64_t* k0col, const int32_t* k1col,
                        int64_t n, uint32_t P, uint16_t* bid, uint32_t* H,
                        int64_t* out) {
    extern __shared__ uint32_t lhist[];
    for (uint32_t x = threadIdx.x; x < P; x += blockDim.x) lhist[x] = 0;
    __syncthreads();
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
            if (!pass) { bid[j] = 0xFFFFu; continue; }
            uint64_t e0 = (uint64_t)k0col[j] ^ 0x8000000000000000ull;
            uint64_t e1 = (uint64_t)(uint32_t)k1col[j];
            uint64_t kk0 = e0 | (e1 << 40);
            uint64_t h = mix64(kk0 ^ 0x9E3779B97F4A7C15ull);
            h = mix64(h);
            uint32_t bk = (uint32_t)(h >> 44) & (P - 1u);
            bid[j] = (uint16_t)bk;
            atomicAdd(&lhist[bk], 1u);
            acc++;
        }
    }
    __syncthreads();
    for (uint32_t x = threadIdx.x; x < P; x += blockDim.x)
        H[(size_t)blockIdx.x * P + x] = lhist[x];
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if ((threadIdx.x & 63) == 0) atomicAdd((unsigned long long*)out, (unsigned long long)acc);
}

/* P4: predk + per-row volatile LDS mode check (the hot-path warmup gate) */
template <int R>
__global__ void k_predm(const int64_t* a, const int64_t* b, const int64_t* c,
                        const int64_t* k0col, const int32_t* k1col,
                        int64_t n, uint32_t P, uint16_t* bid, uint32_t* H,
                        int64_t* out) {
    extern __shared__ uint32_t lhist[];
    volatile uint32_t* lmode = &lhist[P];
    uint32_t* lctr = &lhist[P + 1];
    for (uint32_t x = threadIdx.x; x < P + 3; x += blockDim.x) lhist[x] = 0;
    __syncthreads();
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
            if (!pass) { bid[j] = 0xFFFFu; continue; }
            uint64_t e0 = (uint64_t)k0col[j] ^ 0x8000000000000000ull;
            uint64_t e1 = (uint64_t)(uint32_t)k1col[j];
            uint64_t kk0 = e0 | (e1 << 40);
            /* warmup gate: volatile LDS load + counter bump while mode==0 */
            uint32_t mode = lmode[0];
            if (mode != 1u) {
                uint32_t att = atomicAdd(&lctr[0], 1u);
                if (att == 4095u) lmode[0] = 1u;
            }
            uint64_t h = mix64(kk0 ^ 0x9E3779B97F4A7C15ull);
            h = mix64(h);
            uint32_t bk = (uint32_t)(h >> 44) & (P - 1u);
            bid[j] = (uint16_t)bk;
            atomicAdd(&lhist[bk], 1u);
            acc++;
        }
    }
    __syncthreads();
    for (uint32_t x = threadIdx.x; x < P; x += blockDim.x)
        H[(size_t)blockIdx.x * P + x] = lhist[x];
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if ((threadIdx.x & 63) == 0) atomicAdd((unsigned long long*)out, (unsigned long long)acc);
}

/* P5: predk through the generic spec machinery (runtime col indices in a
 * by-value struct, validity/type/op branches) — models the real row_passes/
 * pack_group_keys codegen. */
struct PCol { int32_t type; const void* data; const uint8_t* valid; };
struct PSpec {
    PCol c[16];
    int32_t n_conj;
    int32_t ccol[8]; int32_t cop[8]; int32_t ctype[8]; int64_t clit[8];
    int32_t n_group; int32_t gcol[4]; int32_t gbits[4]; int64_t gbase[4];
};
template <int R>
__global__ void k_preds(PSpec q, int64_t n, uint32_t P, uint16_t* bid,
                        uint32_t* H, int64_t* out) {
    extern __shared__ uint32_t lhist[];
    for (uint32_t x = threadIdx.x; x < P; x += blockDim.x) lhist[x] = 0;
    __syncthreads();
    int64_t acc = 0;
    int64_t stride = (int64_t)gridDim.x * blockDim.x * R;
    for (int64_t base = ((int64_t)blockIdx.x * blockDim.x) * R; base < n;
         base += stride) {
        int64_t i = base + threadIdx.x;
        #pragma unroll
        for (int k = 0; k < R; k++) {
            int64_t j = i + (int64_t)k * blockDim.x;
            if (j >= n) continue;
            bool pass_all = true;
            #define PEV(J)                 int64_t v##J = 0; bool ok##J = true;                 if (q.n_conj > (J)) {                     const PCol& c = q.c[q.ccol[J]];                     ok##J = c.valid == nullptr || c.valid[j];                     v##J = c.type == 2 ? (int64_t)((const int32_t*)c.data)[j]                                        : ((const int64_t*)c.data)[j];                 }
            PEV(0) PEV(1) PEV(2) PEV(3)
            #undef PEV
            #define PTS(J, V, OK)                 if (q.n_conj > (J)) {                     int cmp = ((V) > q.clit[J]) - ((V) < q.clit[J]);                     bool pass;                     switch (q.cop[J]) {                         case 0: pass = cmp == 0; break;                         case 1: pass = cmp != 0; break;                         case 4: pass = cmp < 0; break;                         default: pass = cmp <= 0; break;                     }                     pass_all = pass_all && (OK) && pass;                 }
            PTS(0, v0, ok0) PTS(1, v1, ok1) PTS(2, v2, ok2) PTS(3, v3, ok3)
            #undef PTS
            if (!pass_all) { bid[j] = 0xFFFFu; continue; }
            uint64_t k0 = 0, k1 = 0; uint32_t flag = 0;
            int shift = 0, word = 0;
            #pragma unroll
            for (int32_t g = 0; g < 4; g++) {
                if (g >= q.n_group) break;
                int bits = q.gbits[g] ? q.gbits[g] : 64;
                if (shift + bits > 64) { word++; shift = 0; }
                const PCol& c = q.c[q.gcol[g]];
                uint64_t e = 0;
                if (!(c.valid == nullptr || c.valid[j])) {
                    flag |= 0x80u >> g;
                } else {
                    e = c.type == 2 ? (uint64_t)(uint32_t)((const int32_t*)c.data)[j]
                        : ((uint64_t)((const int64_t*)c.data)[j]
                           ^ 0x8000000000000000ull);
                    if (bits < 64) e = (e - (uint64_t)q.gbase[g]) &
                                       ((1ull << bits) - 1);
                }
                if (word == 0) k0 |= e << shift; else k1 |= e << shift;
                shift += bits;
            }
            uint64_t h = mix64(k0 ^ 0x9E3779B97F4A7C15ull);
            h = mix64(h ^ k1);
            h = mix64(h ^ flag);
            uint32_t bk = (uint32_t)(h >> 44) & (P - 1u);
            bid[j] = (uint16_t)bk;
            atomicAdd(&lhist[bk], 1u);
            acc++;
        }
    }
    __syncthreads();
    for (uint32_t x = threadIdx.x; x < P; x += blockDim.x)
        H[(size_t)blockIdx.x * P + x] = lhist[x];
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

    int32_t* di; uint32_t* hh;
    CHK(hipMalloc(&di, N * 4));
    CHK(hipMalloc(&hh, (size_t)8192 * 4096 * 4));
    hipLaunchKernelGGL(k_filli32, dim3(4096), dim3(256), 0, 0, di, N, 5);
    CHK(hipDeviceSynchronize());
    static int64_t *A, *B, *C, *D, *OUT; static uint16_t* BID; static int64_t NN;
    static int32_t* DI; static uint32_t* HH; static PSpec QS;
    A = a; B = b; C = c; D = d; OUT = out; BID = bid; NN = N; DI = di; HH = hh;
    QS = PSpec{};
    QS.c[0] = {1, a, nullptr}; QS.c[1] = {1, b, nullptr};
    QS.c[2] = {1, c, nullptr}; QS.c[3] = {1, d, nullptr};
    QS.c[4] = {2, di, nullptr};
    QS.n_conj = 3;
    QS.ccol[0] = 0; QS.cop[0] = 4; QS.clit[0] = 1ll << 30;
    QS.ccol[1] = 1; QS.cop[1] = 4; QS.clit[1] = (int64_t)((1u << 31) * 0.9);
    QS.ccol[2] = 2; QS.cop[2] = 1; QS.clit[2] = 63;
    QS.n_group = 2;
    QS.gcol[0] = 3; QS.gbits[0] = 40; QS.gbase[0] = 0;
    QS.gcol[1] = 4; QS.gbits[1] = 24; QS.gbase[1] = 0;

    struct Case { const char* name; void (*fn)(int, int); double gb; };
    #define L(kern, R, ...) +[](int g, int t) { \
        hipLaunchKernelGGL(kern<R>, dim3(g), dim3(t), 0, 0, __VA_ARGS__); }
    #define LH(kern, R, ...) +[](int g, int t) { \
        hipLaunchKernelGGL(kern<R>, dim3(g), dim3(t), 4096 * 4, 0, __VA_ARGS__); }
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
        {"predh R1", LH(k_predh, 1, A, B, C, NN, 4096u, BID, HH, OUT), 26.0},
        {"predh R2", LH(k_predh, 2, A, B, C, NN, 4096u, BID, HH, OUT), 26.0},
        {"predk R1", LH(k_predk, 1, A, B, C, D, DI, NN, 4096u, BID, HH, OUT), 31.3},
        {"predk R2", LH(k_predk, 2, A, B, C, D, DI, NN, 4096u, BID, HH, OUT), 31.3},
        {"predm R1", LH(k_predm, 1, A, B, C, D, DI, NN, 4096u, BID, HH, OUT), 31.3},
        {"predm R2", LH(k_predm, 2, A, B, C, D, DI, NN, 4096u, BID, HH, OUT), 31.3},
        {"preds R1", LH(k_preds, 1, QS, NN, 4096u, BID, HH, OUT), 31.3},
        {"preds R2", LH(k_preds, 2, QS, NN, 4096u, BID, HH, OUT), 31.3},
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
