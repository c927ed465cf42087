// Microbenchmark: why do the ~1 MB elementwise kernels of the decode
// chain (k_reduce_prep / k_prep_x / finish passes) cost ~5 us?
// Variants isolate: launch floor, grid shape, per-thread depth, reading
// a predecessor's freshly written (dirty-L2) data.
// Build: hipcc -O3 --offload-arch=gfx950 microbench_small_kernel.hip -o mb
#include <hip/hip_runtime.h>
#include <cstdio>

#define CHECK(x) do { auto e = (x); if (e) { printf("ERR %d @%d\n", e, __LINE__); return 1; } } while (0)

// stand-in for a weight-streaming producer (writes a slab, reads a lot)
__global__ void producer(const float* __restrict__ w, float* __restrict__ slab,
                         size_t n, size_t slab_n) {
    float acc = 0.f;
    const size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        acc += w[i];
    const size_t so = (blockIdx.x * blockDim.x + threadIdx.x) % slab_n;
    slab[so] = acc;
}

// the reduce_prep shape: read y (8 floats) + ks slab pairs, write y + half out
__global__ void reducer(float* __restrict__ y, const float* __restrict__ slab,
                        unsigned short* __restrict__ xp, float* __restrict__ ss,
                        int cols, int T, int ks) {
    const int t = blockIdx.x;
    float* yt = y + (size_t)t * cols;
    const int nkc = cols >> 3;
    const int per = (nkc + gridDim.y - 1) / gridDim.y;
    const int kc0 = blockIdx.y * per, kc1 = min(nkc, kc0 + per);
    float sum = 0.f;
    for (int kc = kc0 + threadIdx.x; kc < kc1; kc += blockDim.x) {
        float4 a = *reinterpret_cast<const float4*>(yt + kc * 8);
        float4 b = *reinterpret_cast<const float4*>(yt + kc * 8 + 4);
        for (int k = 0; k < ks; ++k) {
            const float4 p0 = *reinterpret_cast<const float4*>(
                slab + (((size_t)(kc >> 1) * ks + k) * 64 + t) * 16);
            a.x += p0.x; a.y += p0.y; a.z += p0.z; a.w += p0.w;
            b.x += p0.x; b.y += p0.y; b.z += p0.z; b.w += p0.w;
        }
        *reinterpret_cast<float4*>(yt + kc * 8) = a;
        *reinterpret_cast<float4*>(yt + kc * 8 + 4) = b;
        sum += a.x * a.x + b.x * b.x;
        // f16 pack store (8 shorts)
        uint4 o;
        o.x = __float_as_uint(a.x); o.y = __float_as_uint(a.z);
        o.z = __float_as_uint(b.x); o.w = __float_as_uint(b.z);
        *reinterpret_cast<uint4*>(xp + ((size_t)kc * 4 + (t & 3)) * 8) = o;
    }
    if (threadIdx.x == 0) atomicAdd(ss + t, sum);
}

static float time_chain(int iters, bool with_producer, int T, int chunks,
                        float* w, size_t wn, float* slab, size_t slabn,
                        float* y, unsigned short* xp, float* ss, int cols,
                        int ks) {
    hipEvent_t a, b;
    (void)hipEventCreate(&a); (void)hipEventCreate(&b);
    // warmup
    for (int i = 0; i < 3; ++i) {
        if (with_producer)
            hipLaunchKernelGGL(producer, dim3(800), dim3(256), 0, 0, w, slab,
                               wn, slabn);
        hipLaunchKernelGGL(reducer, dim3(T, chunks), dim3(256), 0, 0, y,
                           slab, xp, ss, cols, T, ks);
    }
    (void)hipEventRecord(a);
    for (int i = 0; i < iters; ++i) {
        if (with_producer)
            hipLaunchKernelGGL(producer, dim3(800), dim3(256), 0, 0, w, slab,
                               wn, slabn);
        hipLaunchKernelGGL(reducer, dim3(T, chunks), dim3(256), 0, 0, y,
                           slab, xp, ss, cols, T, ks);
    }
    (void)hipEventRecord(b);
    (void)hipEventSynchronize(b);
    float ms;
    (void)hipEventElapsedTime(&ms, a, b);
    return ms * 1000.f / iters;  // us per iteration (producer+reducer)
}

int main() {
    const int cols = 3200, T = 64, ks = 4;
    const size_t wn = 8u << 20;        // 32 MB of "weights"
    const size_t slabn = (size_t)(cols / 16) * ks * 64 * 16;
    float *w, *slab, *y, *ss;
    unsigned short* xp;
    CHECK(hipMalloc(&w, wn * 4));
    CHECK(hipMalloc(&slab, slabn * 4));
    CHECK(hipMalloc(&y, (size_t)T * cols * 4));
    CHECK(hipMalloc(&xp, (size_t)T * cols * 2 * 4));
    CHECK(hipMalloc(&ss, T * 4));
    CHECK(hipMemset(w, 0, wn * 4));
    CHECK(hipMemset(slab, 0, slabn * 4));
    CHECK(hipMemset(y, 0, (size_t)T * cols * 4));

    printf("reducer alone, grid (%d, C):\n", T);
    for (int chunks : {1, 2, 4, 6, 8, 16})
        printf("  C=%2d: %7.2f us\n", chunks,
               time_chain(200, false, T, chunks, w, wn, slab, slabn, y, xp,
                          ss, cols, ks));
    printf("producer(32MB stream)+reducer chain:\n");
    for (int chunks : {4, 6, 8}) {
        float both = time_chain(100, true, T, chunks, w, wn, slab, slabn, y,
                                xp, ss, cols, ks);
        float prod = time_chain(100, true, T, 1, w, wn, slab, slabn, y, xp,
                                ss, 64, 1);  // tiny reducer ~ producer only
        printf("  C=%2d: chain=%7.2f us (producer-ish=%7.2f)\n", chunks,
               both, prod);
    }
    return 0;
}
