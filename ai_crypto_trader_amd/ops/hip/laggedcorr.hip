// Batched lagged correlation (the social lead/lag
// analysis of social_metrics_analyzer.py:321-455 — Pearson over lags
// -L..+L in one launch, one block per lag). Spearman = the same kernel on
// host-rank-transformed series (ops/social_corr.py).
//
// Pairing convention (== SocialMetricsAnalyzer.lead_lag): lag k >= 0
// pairs a[i] with b[i+k] (a leads b by k); negative k pairs a[i-k], b[i].

#include "common.hpp"

namespace {

__global__ void __launch_bounds__(256) lagged_corr_kernel(
    const float* __restrict__ a,    // (n,)
    const float* __restrict__ b,    // (n,)
    float* __restrict__ out,        // (2*max_lag+1,) pearson per lag
    int n, int max_lag)
{
    __shared__ float scratch[4];    // block_reduce scratch (256/64 waves)
    const int lag = (int)blockIdx.x - max_lag;
    const int m = n - (lag >= 0 ? lag : -lag);
    if (m < 3) {
        if (threadIdx.x == 0) out[blockIdx.x] = 0.0f;
        return;
    }
    const float* xa = lag >= 0 ? a : a - lag;
    const float* xb = lag >= 0 ? b + lag : b;

    float sx = 0.f, sy = 0.f, sxy = 0.f, sxx = 0.f, syy = 0.f;
    for (int i = threadIdx.x; i < m; i += blockDim.x) {
        const float x = xa[i];
        const float y = xb[i];
        sx += x;
        sy += y;
        sxy += x * y;
        sxx += x * x;
        syy += y * y;
    }
    auto add = [] __device__(float p, float q) { return p + q; };
    sx = block_reduce(sx, scratch, add, 0.0f);
    __syncthreads();
    sy = block_reduce(sy, scratch, add, 0.0f);
    __syncthreads();
    sxy = block_reduce(sxy, scratch, add, 0.0f);
    __syncthreads();
    sxx = block_reduce(sxx, scratch, add, 0.0f);
    __syncthreads();
    syy = block_reduce(syy, scratch, add, 0.0f);

    if (threadIdx.x == 0) {
        const float fm = (float)m;
        const float cov = fm * sxy - sx * sy;
        const float vx = fm * sxx - sx * sx;
        const float vy = fm * syy - sy * sy;
        const float den = sqrtf(fmaxf(vx, 0.0f)) * sqrtf(fmaxf(vy, 0.0f));
        out[blockIdx.x] = den > 1e-12f ? cov / den : 0.0f;
    }
}

}  // namespace

extern "C" void launch_lagged_corr(const float* a, const float* b,
                                   float* out, int n, int max_lag,
                                   hipStream_t stream) {
    hipLaunchKernelGGL(lagged_corr_kernel, dim3(2 * max_lag + 1), dim3(256),
                       0, stream, a, b, out, n, max_lag);
}
