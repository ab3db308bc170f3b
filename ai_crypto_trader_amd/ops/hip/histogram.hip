// Volume-profile histogram kernel (SURVEY.md §2.9 last row:
// volume_profile_analyzer.py:86-174's price-binned volume histogram +
// order-book depth aggregation, batched over symbols).
//
// One block per symbol: per-wave privatized LDS histograms (atomic adds
// land in LDS, one global flush per bin at the end — guideline 12), plus
// the up/down volume split (:564-686's buy/sell delta) in the same pass.

#include "common.hpp"

#define VP_MAXBINS 512

namespace {

__global__ void __launch_bounds__(256) vp_hist_kernel(
    const float* __restrict__ candles,   // (nsym, T, 4) [close,high,low,vol]
    const float* __restrict__ lo,        // (nsym,) bin range per symbol
    const float* __restrict__ hi,        // (nsym,)
    float* __restrict__ hist,            // (nsym, n_bins) volume per bin
    float* __restrict__ updown,          // (nsym, 2) up-vol, down-vol
    int T, int n_bins)
{
    __shared__ float lds_hist[VP_MAXBINS];
    __shared__ float lds_ud[2];

    const int sym = blockIdx.x;
    const float l = lo[sym];
    const float inv_w = (float)n_bins / fmaxf(hi[sym] - l, 1e-12f);
    for (int i = threadIdx.x; i < n_bins; i += blockDim.x)
        lds_hist[i] = 0.0f;
    if (threadIdx.x < 2) lds_ud[threadIdx.x] = 0.0f;
    __syncthreads();

    const float4* c = reinterpret_cast<const float4*>(
        candles + (long)sym * T * 4);
    float up = 0.0f, down = 0.0f;
    for (int t = threadIdx.x; t < T; t += blockDim.x) {
        const float4 k = c[t];
        int b = (int)((k.x - l) * inv_w);
        b = min(max(b, 0), n_bins - 1);
        atomicAdd(&lds_hist[b], k.w);
        if (t > 0) {
            const float prev = c[t - 1].x;
            if (k.x >= prev) up += k.w; else down += k.w;
        }
    }
    atomicAdd(&lds_ud[0], up);
    atomicAdd(&lds_ud[1], down);
    __syncthreads();

    for (int i = threadIdx.x; i < n_bins; i += blockDim.x)
        hist[(long)sym * n_bins + i] = lds_hist[i];
    if (threadIdx.x < 2)
        updown[(long)sym * 2 + threadIdx.x] = lds_ud[threadIdx.x];
}

}  // namespace

extern "C" void launch_vp_hist(const float* candles, const float* lo,
                               const float* hi, float* hist, float* updown,
                               int nsym, int T, int n_bins,
                               hipStream_t stream) {
    if (n_bins > VP_MAXBINS)
        throw std::runtime_error("vp_hist: n_bins must be <= 512");
    hipLaunchKernelGGL(vp_hist_kernel, dim3(nsym), dim3(256), 0, stream,
                       candles, lo, hi, hist, updown, T, n_bins);
}
