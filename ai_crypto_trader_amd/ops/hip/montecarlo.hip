// Monte-Carlo correlated-GBM path generator + per-path risk stats
// (replaces monte_carlo_service.py:264-336).
//
// One lane = one portfolio path; each lane carries the A per-asset
// log-prices as a statically-unrolled register array and marches steps:
//   z ~ Philox->Box-Muller (counter-based: reproducible, no state)
//   logS[a] += drift[a] + sum_k cvol[k][a] * z[k]   (cvol = vol*L*sqrt(dt),
//                                                    L = Cholesky factor)
//   V_t = sum_a w[a] * exp(logS[a])  -> running max -> max drawdown
// cvol is staged in LDS ([k][a] row-major, float4-broadcast reads: every
// lane reads the same address -> conflict-free broadcast), so the inner
// correlation loop is pure FMA on registers.
//
// The reference's loop is time-major numpy over 1000 paths
// (monte_carlo_service.py:264-273); here paths are lanes and 10M+ paths
// shard data-parallel across GPUs with an RCCL all-reduce of the final
// stats (parallel/dist.py). Percentiles/VaR/CVaR run on-device on the
// returned per-path vectors.

#include "common.hpp"

namespace {

template <int A>
__global__ void __launch_bounds__(256) mc_paths_kernel(
    const float* __restrict__ cvol,     // (A, A) [k][a] = vol_a*L[a][k]*sqrt(dt)
    const float* __restrict__ drift,    // (A,)   (mu_a - sigma_a^2/2) * dt
    const float* __restrict__ wS0,      // (A,)   weight_a * S0_a
    float* __restrict__ final_value,    // (n_paths,)
    float* __restrict__ max_dd,         // (n_paths,)
    int n_steps, long n_paths, float v0, uint64_t seed)
{
    __shared__ float lds_cvol[A * A];
    __shared__ float lds_drift[A];
    __shared__ float lds_wS0[A];
    for (int i = threadIdx.x; i < A * A; i += blockDim.x)
        lds_cvol[i] = cvol[i];
    for (int i = threadIdx.x; i < A; i += blockDim.x) {
        lds_drift[i] = drift[i];
        lds_wS0[i] = wS0[i];
    }
    __syncthreads();

    const long stride = (long)gridDim.x * blockDim.x;
    for (long path = (long)blockIdx.x * blockDim.x + threadIdx.x;
         path < n_paths; path += stride) {
        float logS[A];
#pragma unroll
        for (int a = 0; a < A; ++a) logS[a] = 0.0f;
        float vmax = v0, mdd = 0.0f, V = v0;

        for (int step = 0; step < n_steps; ++step) {
            for (int k4 = 0; k4 < A / 4; ++k4) {
                float4 z4 = philox_normal4(
                    seed, (uint64_t)path,
                    ((uint64_t)step << 32) | (uint64_t)k4);
#pragma unroll
                for (int dz = 0; dz < 4; ++dz) {
                    const float z = dz == 0   ? z4.x
                                    : dz == 1 ? z4.y
                                    : dz == 2 ? z4.z
                                              : z4.w;
                    const int k = k4 * 4 + dz;
                    const float4* row =
                        reinterpret_cast<const float4*>(lds_cvol + k * A);
#pragma unroll
                    for (int a4 = 0; a4 < A / 4; ++a4) {
                        float4 cv = row[a4];
                        logS[a4 * 4 + 0] += cv.x * z;
                        logS[a4 * 4 + 1] += cv.y * z;
                        logS[a4 * 4 + 2] += cv.z * z;
                        logS[a4 * 4 + 3] += cv.w * z;
                    }
                }
            }
            V = 0.0f;
#pragma unroll
            for (int a = 0; a < A; ++a) {
                logS[a] += lds_drift[a];
                V += lds_wS0[a] * __expf(logS[a]);
            }
            vmax = fmaxf(vmax, V);
            mdd = fmaxf(mdd, (vmax - V) / vmax);
        }
        final_value[path] = V;
        max_dd[path] = mdd;
    }
}

}  // namespace

extern "C" void launch_mc_paths(const float* cvol, const float* drift,
                                const float* wS0, const float* weights_unused,
                                float* final_value, float* max_dd,
                                int n_assets, int n_steps, long n_paths,
                                float v0, uint64_t seed, hipStream_t stream) {
    (void)weights_unused;
    long want = (n_paths + 255) / 256;
    int blocks = (int)(want < 8192 ? want : 8192);
    dim3 grid(blocks), block(256);
#define MC_CASE(AA)                                                           \
    case AA:                                                                  \
        hipLaunchKernelGGL(mc_paths_kernel<AA>, grid, block, 0, stream, cvol, \
                           drift, wS0, final_value, max_dd, n_steps, n_paths, \
                           v0, seed);                                         \
        break;
    switch (n_assets) {
        MC_CASE(4)
        MC_CASE(8)
        MC_CASE(16)
        MC_CASE(32)
        MC_CASE(64)
        default:
            throw std::runtime_error(
                "mc_paths: n_assets must be one of 4/8/16/32/64");
    }
#undef MC_CASE
}
