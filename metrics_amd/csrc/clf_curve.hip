// K2 — exact classification-curve core for gfx950 (CDNA4).
//
// Implements the reference's `_binary_clf_curve`
// (torchmetrics functional/classification/precision_recall_curve.py:30-82):
// descending sort of the scores, cumulative tp/fp at each DISTINCT score.
// MI355X-native formulation: rocPRIM device radix sort (stable, fp32 keys)
// + double-precision device scans + a fused gather/flag kernel and a fused
// compaction kernel, all launched on the caller's torch HIP stream.
// Counts are integers (< 2^53), so the fp64 scan is bit-exact for the
// unweighted path and at least as accurate as the reference's fp32 cumsum
// for the weighted one.
//
// Exposed C API (ctypes, see ops/_hip.py):
//   ma_clf_curve_scratch_bytes(N, weighted, &bytes)
//   ma_binary_clf_curve(stream, preds, target, weight|0, N, pos_label,
//                       scratch, scratch_bytes, out_fps, out_tps,
//                       out_thresh, out_count)

#include <cstring>  // memset, needed by rocprim's texture_cache_iterator header

#include <hip/hip_runtime.h>
#include <rocprim/rocprim.hpp>

#include <cstdint>

using ll = long long;

#define OK(x)                                                                  \
    do {                                                                       \
        hipError_t err_ = (x);                                                 \
        if (err_ != hipSuccess) return (int)err_;                              \
    } while (0)

static constexpr size_t ALIGN = 256;

static inline size_t align_up(size_t x) { return (x + ALIGN - 1) & ~(ALIGN - 1); }

// ---------------------------------------------------------------- kernels

__global__ void __launch_bounds__(256) k_iota(int* idx, ll n) {
    ll i = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) idx[i] = (int)i;
}

// Gather target/weight through the sort permutation and mark distinct-score
// boundaries in one pass.
__global__ void __launch_bounds__(256) k_gather_flags(
    const float* __restrict__ keys_sorted, const int* __restrict__ idx_sorted,
    const ll* __restrict__ target, const float* __restrict__ weight, ll n,
    ll pos_label, double* __restrict__ tvals, double* __restrict__ fvals,
    int* __restrict__ flags) {
    ll i = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const int src = idx_sorted[i];
    const double t = (target[src] == pos_label) ? 1.0 : 0.0;
    const double w = weight ? (double)weight[src] : 1.0;
    tvals[i] = t * w;
    if (fvals) fvals[i] = (1.0 - t) * w;
    flags[i] = (i == n - 1) || (keys_sorted[i] != keys_sorted[i + 1]);
}

// Compact (tps, fps, thresholds) at the flagged positions. pos_scan is the
// INCLUSIVE scan of flags, so out position = pos_scan[i] - 1.
__global__ void __launch_bounds__(256) k_compact(
    const float* __restrict__ keys_sorted, const int* __restrict__ flags,
    const int* __restrict__ pos_scan, const double* __restrict__ tps_full,
    const double* __restrict__ fps_full /* null => unweighted */, ll n,
    float* __restrict__ out_fps, float* __restrict__ out_tps,
    float* __restrict__ out_thresh, ll* __restrict__ out_count) {
    ll i = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (flags[i]) {
        const int o = pos_scan[i] - 1;
        const double tp = tps_full[i];
        out_tps[o] = (float)tp;
        out_fps[o] = fps_full ? (float)fps_full[i] : (float)((double)(i + 1) - tp);
        out_thresh[o] = keys_sorted[i];
    }
    if (i == n - 1) *out_count = (ll)pos_scan[n - 1];
}

// ----------------------------------------------------------------- C API

extern "C" int ma_clf_curve_scratch_bytes(ll n, int weighted, unsigned long long* out_bytes) {
    size_t sort_tmp = 0, scan_tmp = 0, iscan_tmp = 0;
    hipError_t e;
    e = rocprim::radix_sort_pairs_desc((void*)nullptr, sort_tmp, (const float*)nullptr,
                                       (float*)nullptr, (const int*)nullptr, (int*)nullptr,
                                       (size_t)n);
    if (e != hipSuccess) return (int)e;
    e = rocprim::inclusive_scan((void*)nullptr, scan_tmp, (const double*)nullptr,
                                (double*)nullptr, (size_t)n, rocprim::plus<double>());
    if (e != hipSuccess) return (int)e;
    e = rocprim::inclusive_scan((void*)nullptr, iscan_tmp, (const int*)nullptr,
                                (int*)nullptr, (size_t)n, rocprim::plus<int>());
    if (e != hipSuccess) return (int)e;
    size_t tmp = sort_tmp;
    if (scan_tmp > tmp) tmp = scan_tmp;
    if (iscan_tmp > tmp) tmp = iscan_tmp;

    size_t total = 0;
    total += align_up((size_t)n * sizeof(float));   // keys_sorted
    total += align_up((size_t)n * sizeof(int));     // idx (iota)
    total += align_up((size_t)n * sizeof(int));     // idx_sorted
    total += align_up((size_t)n * sizeof(double));  // tvals / tps_full (in-place scan)
    if (weighted) total += align_up((size_t)n * sizeof(double));  // fvals / fps_full
    total += align_up((size_t)n * sizeof(int));     // flags
    total += align_up((size_t)n * sizeof(int));     // pos_scan
    total += align_up(tmp);                          // rocprim temp
    *out_bytes = (unsigned long long)total;
    return 0;
}

extern "C" int ma_binary_clf_curve(
    uint64_t stream_u, uint64_t preds_u, uint64_t target_u, uint64_t weight_u, ll n,
    ll pos_label, uint64_t scratch_u, unsigned long long scratch_bytes,
    uint64_t out_fps_u, uint64_t out_tps_u, uint64_t out_thresh_u, uint64_t out_count_u) {
    hipStream_t stream = (hipStream_t)stream_u;
    const float* preds = (const float*)preds_u;
    const ll* target = (const ll*)target_u;
    const float* weight = (const float*)weight_u;  // may be null

    char* p = (char*)scratch_u;
    float* keys_sorted = (float*)p;          p += align_up((size_t)n * sizeof(float));
    int* idx = (int*)p;                      p += align_up((size_t)n * sizeof(int));
    int* idx_sorted = (int*)p;               p += align_up((size_t)n * sizeof(int));
    double* tvals = (double*)p;              p += align_up((size_t)n * sizeof(double));
    double* fvals = nullptr;
    if (weight) { fvals = (double*)p;        p += align_up((size_t)n * sizeof(double)); }
    int* flags = (int*)p;                    p += align_up((size_t)n * sizeof(int));
    int* pos_scan = (int*)p;                 p += align_up((size_t)n * sizeof(int));
    void* rp_tmp = (void*)p;
    size_t rp_bytes = (size_t)((char*)scratch_u + scratch_bytes - p);

    const int B = 256;
    const ll grid = (n + B - 1) / B;

    hipLaunchKernelGGL(k_iota, dim3(grid), dim3(B), 0, stream, idx, n);
    size_t tmp_bytes = rp_bytes;
    OK(rocprim::radix_sort_pairs_desc(rp_tmp, tmp_bytes, preds, keys_sorted, idx, idx_sorted,
                                      (size_t)n, 0, 32, stream));
    hipLaunchKernelGGL(k_gather_flags, dim3(grid), dim3(B), 0, stream, keys_sorted, idx_sorted,
                       target, weight, n, pos_label, tvals, fvals, flags);
    tmp_bytes = rp_bytes;
    OK(rocprim::inclusive_scan(rp_tmp, tmp_bytes, tvals, tvals, (size_t)n,
                               rocprim::plus<double>(), stream));
    if (fvals) {
        tmp_bytes = rp_bytes;
        OK(rocprim::inclusive_scan(rp_tmp, tmp_bytes, fvals, fvals, (size_t)n,
                                   rocprim::plus<double>(), stream));
    }
    tmp_bytes = rp_bytes;
    OK(rocprim::inclusive_scan(rp_tmp, tmp_bytes, flags, pos_scan, (size_t)n,
                               rocprim::plus<int>(), stream));
    hipLaunchKernelGGL(k_compact, dim3(grid), dim3(B), 0, stream, keys_sorted, flags, pos_scan,
                       tvals, fvals, n, (float*)out_fps_u, (float*)out_tps_u,
                       (float*)out_thresh_u, (ll*)out_count_u);
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------------
// Retrieval grouping sort: ONE radix pass replaces the torch lexsort
// (argsort pred desc -> argsort index asc) via a composite 64-bit key
// (query index << 32) | descending-order-preserving score bits; a second
// 32-bit sort yields the by-index (stable original order) permutation.

__global__ void __launch_bounds__(256) k_retrieval_keys(
    const ll* __restrict__ idx, const float* __restrict__ preds, ll n,
    unsigned long long* __restrict__ keys, unsigned int* __restrict__ ikeys, int* __restrict__ pos) {
    ll i = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    unsigned int b = __float_as_uint(preds[i]);
    // ascending-order-preserving flip, then invert for descending score
    b = (b & 0x80000000u) ? ~b : (b | 0x80000000u);
    b = ~b;
    keys[i] = ((unsigned long long)(unsigned int)idx[i] << 32) | (unsigned long long)b;
    ikeys[i] = (unsigned int)idx[i];
    pos[i] = (int)i;
}

extern "C" int ma_retrieval_sort_scratch_bytes(ll n, unsigned long long* out_bytes) {
    size_t s64 = 0, s32 = 0;
    hipError_t e;
    e = rocprim::radix_sort_pairs((void*)nullptr, s64, (const unsigned long long*)nullptr,
                                  (unsigned long long*)nullptr, (const int*)nullptr, (int*)nullptr,
                                  (size_t)n);
    if (e != hipSuccess) return (int)e;
    e = rocprim::radix_sort_pairs((void*)nullptr, s32, (const unsigned int*)nullptr,
                                  (unsigned int*)nullptr, (const int*)nullptr, (int*)nullptr,
                                  (size_t)n);
    if (e != hipSuccess) return (int)e;
    size_t tmp = s64 > s32 ? s64 : s32;
    size_t total = 0;
    total += align_up((size_t)n * sizeof(unsigned long long));  // keys in
    total += align_up((size_t)n * sizeof(unsigned long long));  // keys out
    total += align_up((size_t)n * sizeof(unsigned int));        // ikeys in
    total += align_up((size_t)n * sizeof(unsigned int));        // ikeys out
    total += align_up((size_t)n * sizeof(int));                 // pos
    total += align_up(tmp);
    *out_bytes = (unsigned long long)total;
    return 0;
}

extern "C" int ma_retrieval_sort(
    uint64_t stream_u, uint64_t idx_u, uint64_t preds_u, ll n,
    uint64_t scratch_u, unsigned long long scratch_bytes,
    uint64_t out_order_u, uint64_t out_by_index_u) {
    hipStream_t stream = (hipStream_t)stream_u;
    char* p = (char*)scratch_u;
    unsigned long long* keys_in = (unsigned long long*)p;  p += align_up((size_t)n * sizeof(unsigned long long));
    unsigned long long* keys_out = (unsigned long long*)p; p += align_up((size_t)n * sizeof(unsigned long long));
    unsigned int* ikeys_in = (unsigned int*)p;             p += align_up((size_t)n * sizeof(unsigned int));
    unsigned int* ikeys_out = (unsigned int*)p;            p += align_up((size_t)n * sizeof(unsigned int));
    int* pos = (int*)p;                                    p += align_up((size_t)n * sizeof(int));
    void* tmp = (void*)p;
    size_t tmp_bytes = (size_t)((char*)scratch_u + scratch_bytes - p);

    const ll grid = (n + 255) / 256;
    hipLaunchKernelGGL(k_retrieval_keys, dim3(grid), dim3(256), 0, stream,
                       (const ll*)idx_u, (const float*)preds_u, n, keys_in, ikeys_in, pos);
    size_t b = tmp_bytes;
    OK(rocprim::radix_sort_pairs(tmp, b, keys_in, keys_out, pos, (int*)out_order_u,
                                 (size_t)n, 0, 64, stream));
    b = tmp_bytes;
    OK(rocprim::radix_sort_pairs(tmp, b, ikeys_in, ikeys_out, pos, (int*)out_by_index_u,
                                 (size_t)n, 0, 32, stream));
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------------
// Batched multiclass/multilabel exact curves: ONE composite-key sort for all
// classes ((class << 32) | descending-score bits) replaces C per-class sorts,
// and ONE (C,) count transfer replaces C data-dependent-size syncs. Each
// class's segment is exactly B elements, so segment bases are implicit.

__global__ void __launch_bounds__(256) k_mc_keys(
    const float* __restrict__ probs /* (B, C) */, ll B, ll C,
    unsigned long long* __restrict__ keys, int* __restrict__ pos) {
    ll e = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (e >= B * C) return;
    const ll c = e % C;  // element e = (row, c) row-major
    unsigned int b = __float_as_uint(probs[e]);
    b = (b & 0x80000000u) ? ~b : (b | 0x80000000u);
    b = ~b;  // descending within class
    keys[e] = ((unsigned long long)c << 32) | (unsigned long long)b;
    pos[e] = (int)e;
}

__global__ void __launch_bounds__(256) k_mc_gather_flags(
    const unsigned long long* __restrict__ keys_sorted, const int* __restrict__ pos_sorted,
    const float* __restrict__ probs, const ll* __restrict__ target, ll B, ll C, ll n,
    double* __restrict__ tvals, int* __restrict__ flags) {
    ll i = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const ll src = pos_sorted[i];
    const ll row = src / C;
    const ll cls = (ll)(keys_sorted[i] >> 32);
    tvals[i] = (target[row] == cls) ? 1.0 : 0.0;
    flags[i] = (i == n - 1) || (keys_sorted[i] != keys_sorted[i + 1]);
}

// multilabel variant: target is (B, C) 0/1
__global__ void __launch_bounds__(256) k_ml_gather_flags(
    const unsigned long long* __restrict__ keys_sorted, const int* __restrict__ pos_sorted,
    const ll* __restrict__ target, ll n,
    double* __restrict__ tvals, int* __restrict__ flags) {
    ll i = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    tvals[i] = (double)target[pos_sorted[i]];
    flags[i] = (i == n - 1) || (keys_sorted[i] != keys_sorted[i + 1]);
}

__global__ void __launch_bounds__(256) k_mc_compact(
    const unsigned long long* __restrict__ keys_sorted, const int* __restrict__ pos_sorted,
    const float* __restrict__ probs, const int* __restrict__ flags,
    const int* __restrict__ pos_scan, const double* __restrict__ tps_full, ll B, ll C, ll n,
    float* __restrict__ out_fps, float* __restrict__ out_tps, float* __restrict__ out_thr) {
    ll i = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (!flags[i]) return;
    const ll cls = (ll)(keys_sorted[i] >> 32);
    const ll seg0 = cls * B;
    const double base = (seg0 > 0) ? tps_full[seg0 - 1] : 0.0;
    const double tp = tps_full[i] - base;
    const int o = pos_scan[i] - 1;
    out_tps[o] = (float)tp;
    out_fps[o] = (float)((double)(i - seg0 + 1) - tp);
    out_thr[o] = probs[pos_sorted[i]];
}

__global__ void __launch_bounds__(256) k_mc_counts(
    const int* __restrict__ pos_scan, ll B, ll C, ll* __restrict__ counts) {
    ll c = (ll)blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    const int hi = pos_scan[(c + 1) * B - 1];
    const int lo = (c > 0) ? pos_scan[c * B - 1] : 0;
    counts[c] = hi - lo;
}

extern "C" int ma_mc_clf_curve_scratch_bytes(ll B, ll C, unsigned long long* out_bytes) {
    const ll n = B * C;
    size_t sort_tmp = 0, scan_tmp = 0, iscan_tmp = 0;
    hipError_t e;
    e = rocprim::radix_sort_pairs((void*)nullptr, sort_tmp, (const unsigned long long*)nullptr,
                                  (unsigned long long*)nullptr, (const int*)nullptr, (int*)nullptr,
                                  (size_t)n);
    if (e != hipSuccess) return (int)e;
    e = rocprim::inclusive_scan((void*)nullptr, scan_tmp, (const double*)nullptr,
                                (double*)nullptr, (size_t)n, rocprim::plus<double>());
    if (e != hipSuccess) return (int)e;
    e = rocprim::inclusive_scan((void*)nullptr, iscan_tmp, (const int*)nullptr,
                                (int*)nullptr, (size_t)n, rocprim::plus<int>());
    if (e != hipSuccess) return (int)e;
    size_t tmp = sort_tmp;
    if (scan_tmp > tmp) tmp = scan_tmp;
    if (iscan_tmp > tmp) tmp = iscan_tmp;
    size_t total = 0;
    total += align_up((size_t)n * sizeof(unsigned long long));  // keys in
    total += align_up((size_t)n * sizeof(unsigned long long));  // keys out
    total += align_up((size_t)n * sizeof(int));                 // pos in
    total += align_up((size_t)n * sizeof(int));                 // pos out
    total += align_up((size_t)n * sizeof(double));              // tvals/scan
    total += align_up((size_t)n * sizeof(int));                 // flags
    total += align_up((size_t)n * sizeof(int));                 // pos_scan
    total += align_up(tmp);
    *out_bytes = (unsigned long long)total;
    return 0;
}

extern "C" int ma_mc_clf_curve(
    uint64_t stream_u, uint64_t probs_u, uint64_t target_u, ll B, ll C, int multilabel,
    uint64_t scratch_u, unsigned long long scratch_bytes,
    uint64_t out_fps_u, uint64_t out_tps_u, uint64_t out_thr_u, uint64_t out_counts_u) {
    hipStream_t stream = (hipStream_t)stream_u;
    const ll n = B * C;
    char* p = (char*)scratch_u;
    unsigned long long* keys_in = (unsigned long long*)p;  p += align_up((size_t)n * sizeof(unsigned long long));
    unsigned long long* keys_out = (unsigned long long*)p; p += align_up((size_t)n * sizeof(unsigned long long));
    int* pos_in = (int*)p;                                 p += align_up((size_t)n * sizeof(int));
    int* pos_out = (int*)p;                                p += align_up((size_t)n * sizeof(int));
    double* tvals = (double*)p;                            p += align_up((size_t)n * sizeof(double));
    int* flags = (int*)p;                                  p += align_up((size_t)n * sizeof(int));
    int* pos_scan = (int*)p;                               p += align_up((size_t)n * sizeof(int));
    void* tmp = (void*)p;
    size_t tmp_bytes = (size_t)((char*)scratch_u + scratch_bytes - p);

    const ll grid = (n + 255) / 256;
    hipLaunchKernelGGL(k_mc_keys, dim3(grid), dim3(256), 0, stream,
                       (const float*)probs_u, B, C, keys_in, pos_in);
    size_t b = tmp_bytes;
    OK(rocprim::radix_sort_pairs(tmp, b, keys_in, keys_out, pos_in, pos_out, (size_t)n, 0, 64, stream));
    if (multilabel)
        hipLaunchKernelGGL(k_ml_gather_flags, dim3(grid), dim3(256), 0, stream,
                           keys_out, pos_out, (const ll*)target_u, n, tvals, flags);
    else
        hipLaunchKernelGGL(k_mc_gather_flags, dim3(grid), dim3(256), 0, stream,
                           keys_out, pos_out, (const float*)probs_u, (const ll*)target_u, B, C, n,
                           tvals, flags);
    b = tmp_bytes;
    OK(rocprim::inclusive_scan(tmp, b, tvals, tvals, (size_t)n, rocprim::plus<double>(), stream));
    b = tmp_bytes;
    OK(rocprim::inclusive_scan(tmp, b, flags, pos_scan, (size_t)n, rocprim::plus<int>(), stream));
    hipLaunchKernelGGL(k_mc_compact, dim3(grid), dim3(256), 0, stream,
                       keys_out, pos_out, (const float*)probs_u, flags, pos_scan, tvals, B, C, n,
                       (float*)out_fps_u, (float*)out_tps_u, (float*)out_thr_u);
    hipLaunchKernelGGL(k_mc_counts, dim3((C + 255) / 256), dim3(256), 0, stream,
                       pos_scan, B, C, (ll*)out_counts_u);
    return (int)hipGetLastError();
}
