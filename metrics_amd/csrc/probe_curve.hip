// Ablation probe for k_multiclass_curve_hist: which phase costs 600us?
// Variants: 0=full, 1=no-atomic (keep value alive), 2=load-only, 3=u32-atomic,
// 4=full with positives/negatives split (negatives to per-wave aggregated j==0 path)
#include <hip/hip_runtime.h>
#include <stdint.h>
typedef long long ll;

__device__ __forceinline__ float bf16f(unsigned short u) {
    unsigned int v = ((unsigned int)u) << 16;
    return __uint_as_float(v);
}

__device__ __forceinline__ int buck(float p, const float* thr, int T, float t0, float inv_step) {
    int j = (int)floorf((p - t0) * inv_step) + 1;
    j = j < 0 ? 0 : (j > T ? T : j);
    while (j < T && thr[j] <= p) j++;
    while (j > 0 && thr[j - 1] > p) j--;
    return j;
}

template <int VARIANT>
__global__ void __launch_bounds__(256) probe(
    const unsigned short* __restrict__ probs, const ll* __restrict__ target, ll B, ll C,
    const float* __restrict__ thresholds, int T, float t0, float inv_step,
    unsigned long long* __restrict__ hist, unsigned int* __restrict__ hist32) {
    extern __shared__ float sthr[];
    for (int b = threadIdx.x; b < T; b += blockDim.x) sthr[b] = thresholds[b];
    __syncthreads();
    const unsigned int NC = (unsigned int)C;
    const unsigned long long total = (unsigned long long)B * NC;
    unsigned long long i4 = ((unsigned long long)blockIdx.x * blockDim.x + threadIdx.x) * 4ULL;
    const unsigned long long stride4 = (unsigned long long)gridDim.x * blockDim.x * 4ULL;
    unsigned long long sink = 0;
    for (; i4 < total; i4 += stride4) {
        int nv = (int)(total - i4 < 4 ? total - i4 : 4);
        float pv[4];
        if (nv == 4) {
            ushort4 u = *reinterpret_cast<const ushort4*>(probs + i4);
            pv[0] = bf16f(u.x); pv[1] = bf16f(u.y); pv[2] = bf16f(u.z); pv[3] = bf16f(u.w);
        } else {
            for (int k = 0; k < nv; k++) pv[k] = bf16f(probs[i4 + k]);
        }
        unsigned long long row = i4 / NC;
        unsigned int c = (unsigned int)(i4 - row * NC);
        ll trow = target[row];
        for (int k = 0; k < nv; k++) {
            if (c >= NC) { row++; c = 0; trow = target[row]; }
            int label = (trow == (ll)c) ? 1 : 0;
            if (VARIANT == 2) { sink += (unsigned long long)(pv[k] > 0.5f) + label; c++; continue; }
            int j = buck(pv[k], sthr, T, t0, inv_step);
            unsigned long long addr = ((unsigned long long)c * (T + 1) + j) * 2 + label;
            if (VARIANT == 0) atomicAdd(&hist[addr], 1ULL);
            else if (VARIANT == 1) sink += addr;
            else if (VARIANT == 3) atomicAdd(&hist32[addr], 1u);
            c++;
        }
    }
    if (VARIANT == 1 || VARIANT == 2) {
        asm volatile("" ::"v"(sink));  // keep live (guide rule #17)
        if (sink == 0xdeadbeefULL) hist[0] = sink;
    }
}

extern "C" int probe_curve(uintptr_t stream, int variant, uintptr_t probs, uintptr_t target, ll B,
                           ll C, uintptr_t thr, int T, float t0, float inv_step, uintptr_t hist,
                           uintptr_t hist32, int grid) {
    hipStream_t s = (hipStream_t)stream;
    size_t sh = (size_t)T * sizeof(float);
    switch (variant) {
        case 0: probe<0><<<grid, 256, sh, s>>>((const unsigned short*)probs, (const ll*)target, B, C, (const float*)thr, T, t0, inv_step, (unsigned long long*)hist, (unsigned int*)hist32); break;
        case 1: probe<1><<<grid, 256, sh, s>>>((const unsigned short*)probs, (const ll*)target, B, C, (const float*)thr, T, t0, inv_step, (unsigned long long*)hist, (unsigned int*)hist32); break;
        case 2: probe<2><<<grid, 256, sh, s>>>((const unsigned short*)probs, (const ll*)target, B, C, (const float*)thr, T, t0, inv_step, (unsigned long long*)hist, (unsigned int*)hist32); break;
        case 3: probe<3><<<grid, 256, sh, s>>>((const unsigned short*)probs, (const ll*)target, B, C, (const float*)thr, T, t0, inv_step, (unsigned long long*)hist, (unsigned int*)hist32); break;
    }
    return (int)hipGetLastError();
}
