// Monte-Carlo correlated-GBM path generator + per-path risk stats
// (replaces monte_carlo_service.py:264-336).
//
// One lane = one portfolio path; each lane carries the A per-asset
// log-prices as a statically-unrolled register array and marches steps:
//   z ~ Philox->Box-Muller (counter-based: reproducible, no state)
//   logS[a] += drift[a] + sum_k cvol[k][a] * z[k]   (cvol = vol*L*sqrt(dt),
//                                                    L = Cholesky factor)
//   V_t = sum_a w[a] * exp2(logS[a])  -> running max -> max drawdown
// BASE-2 log space: the wrapper pre-scales drift and cvol by log2(e)
// (ops/montecarlo.py _gbm_terms, shared with the CPU reference), so the
// per-asset-per-step exponential is native v_exp_f32 with no log2e
// multiply — the kernel is VALU-issue-bound and exp runs at 1/4 rate, so
// every instruction on that path counts.
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

typedef __bf16 mc_bf16;
typedef __bf16 mc_bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 mc_bf16x4 __attribute__((ext_vector_type(4)));
typedef float mc_f32x4 __attribute__((ext_vector_type(4)));

namespace {

template <int A>
__global__ void __launch_bounds__(256) mc_paths_kernel(
    const float* __restrict__ cvol,     // (A, A) [k][a] = vol_a*L[a][k]*sqrt(dt)
    const float* __restrict__ drift,    // (A,)   (mu_a - sigma_a^2/2) * dt
    const float* __restrict__ wS0,      // (A,)   weight_a * S0_a
    float* __restrict__ final_value,    // (n_paths,)
    float* __restrict__ max_dd,         // (n_paths,)
    int n_steps, long n_paths, float v0, uint64_t seed, long path_base,
    int antithetic)
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

        // antithetic pairing: the top half of the path range mirrors the
        // bottom half's normals with flipped sign (classic MC variance
        // reduction; counter-based RNG makes the pairing stateless)
        const long half = n_paths / 2;
        const bool mirror = antithetic && path >= half;
        const long zsrc = (mirror ? path - half : path) + path_base;
        const float zsign = mirror ? -1.0f : 1.0f;
        for (int step = 0; step < n_steps; ++step) {
            for (int k4 = 0; k4 < A / 4; ++k4) {
                float4 z4 = philox_normal4(
                    seed, (uint64_t)zsrc,
                    ((uint64_t)step << 32) | (uint64_t)k4);
                z4.x *= zsign; z4.y *= zsign;
                z4.z *= zsign; z4.w *= zsign;
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
                V += lds_wS0[a] * __builtin_amdgcn_exp2f(logS[a]);
            }
            vmax = fmaxf(vmax, V);
            mdd = fmaxf(mdd, (vmax - V) * __builtin_amdgcn_rcpf(vmax));  // 1-ulp rcp: dd is a statistic
        }
        final_value[path] = V;
        max_dd[path] = mdd;
    }
}

// ---------------------------------------------------------------------------
// MFMA variant (A = 64 assets): the correlation step z_corr = CVOL @ Z runs
// on v_mfma_f32_16x16x32_bf16 with the per-path LOG-PRICES KEPT IN THE
// ACCUMULATOR FRAGMENTS across all steps — the MFMA's C-in/C-out chaining
// IS the path recurrence logS += CVOL @ z. Block = 4 waves = 256 paths;
// per step each thread Philox-generates its own path's 64 normals into a
// row of the LDS Z^T tile (16-B-contiguous per thread), wave w then owns
// path-columns [64w, 64w+64) of C = CVOL(64x64) @ Z(64x256).
// Portfolio value per path = cross-fragment reduction over the asset rows:
// per-lane partial over its 16 rows, then shfl-xor over the 4 lanes that
// share a column (l, l+16, l+32, l+48).
//
// bf16 deliberately (NOT fp8): gfx950 non-scaled fp8 MFMA runs at the bf16
// rate (cdna_hip_programming.md §3 table: 2047 vs 2382 TF), so e4m3 would
// buy nothing and cost ~2 bits of Z/L mantissa. Accumulation stays f32.
// The f32 VALU kernel above remains the bit-exact golden reference
// (mc_paths_cpu); this one is validated statistically + against it at
// ~1e-2 tolerance (bf16 quantization of Z and CVOL).
// ---------------------------------------------------------------------------

#define MCM_A 64
#define MCM_PB 256          // paths per block
#define MCM_KP 72           // padded bf16 row stride (16-lane groups spread)

__global__ void __launch_bounds__(512) mc_paths_mfma_kernel(
    const float* __restrict__ cvol,     // (A, A) [k][a] (VALU-kernel layout)
    const float* __restrict__ drift,    // (A,)
    const float* __restrict__ wS0,      // (A,)
    float* __restrict__ final_value,    // (n_paths,)
    float* __restrict__ max_dd,         // (n_paths,)
    int n_steps, long n_paths, float v0, uint64_t seed, long path_base,
    int antithetic)
{
    // 8 waves: wave w owns path-columns [32w, 32w+32) -> 8 accumulator
    // fragments per lane (32 f32) instead of 16, so 2+ waves/SIMD fit;
    // each thread Philox-fills HALF a Z^T row (RNG parallelized 2x).
    __shared__ mc_bf16 lds_m[MCM_A * MCM_KP];        // CVOL, row-major [a][k]
    __shared__ mc_bf16 lds_zt[MCM_PB * MCM_KP];      // Z^T: [path][k]
    __shared__ float lds_drift[MCM_A];
    __shared__ float lds_w[MCM_A];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;             // 8 waves
    const int fr = lane & 15;
    const int fq = lane >> 4;

    for (int i = tid; i < MCM_A * MCM_A; i += 512)
        lds_m[(i / MCM_A) * MCM_KP + (i % MCM_A)] =
            (mc_bf16)cvol[(i % MCM_A) * MCM_A + (i / MCM_A)];
    for (int i = tid; i < MCM_A; i += 512) {
        lds_drift[i] = drift[i];
        lds_w[i] = wS0[i];
    }
    __syncthreads();

    const long path0 = (long)blockIdx.x * MCM_PB;
    if (path0 >= n_paths) return;
    const int zrow = tid & (MCM_PB - 1);      // this thread's Z^T row
    const int zhalf = tid >> 8;               // which 32-normal half
    // antithetic pairing across the global path range (see VALU kernel)
    const long gpath = path0 + zrow;
    const long half = n_paths / 2;
    const bool mirror = antithetic && gpath >= half;
    const long zpath = (mirror ? gpath - half : gpath) + path_base;
    const float zsign = mirror ? -1.0f : 1.0f;

    mc_f32x4 logS[4][2];                      // [m_tile][col_tile]
#pragma unroll
    for (int mt = 0; mt < 4; ++mt)
#pragma unroll
        for (int ct = 0; ct < 2; ++ct) logS[mt][ct] = {0.f, 0.f, 0.f, 0.f};

    // this lane's 16 asset rows never change: hoist drift/weight out of
    // LDS into registers (saves 64 broadcast reads per step per lane)
    float r_drift[4][4], r_w[4][4];
#pragma unroll
    for (int mt = 0; mt < 4; ++mt)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int row = mt * 16 + fq * 4 + r;
            r_drift[mt][r] = lds_drift[row];
            r_w[mt][r] = lds_w[row];
        }

    float vmax[2], mdd[2], V[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) { vmax[c] = v0; mdd[c] = 0.0f; V[c] = v0; }

    for (int step = 0; step < n_steps; ++step) {
#pragma unroll
        for (int j = 0; j < MCM_A / 8; ++j) {
            const int k4 = zhalf * (MCM_A / 8) + j;
            float4 z4 = philox_normal4(
                seed, (uint64_t)zpath,
                ((uint64_t)step << 32) | (uint64_t)k4);
            // one 8-byte store (offset = zrow*144 + k4*8 bytes, 8-aligned)
            *reinterpret_cast<mc_bf16x4*>(lds_zt + zrow * MCM_KP + k4 * 4) =
                mc_bf16x4{(mc_bf16)(z4.x * zsign), (mc_bf16)(z4.y * zsign),
                          (mc_bf16)(z4.z * zsign),
                          (mc_bf16)(z4.w * zsign)};
        }
        __syncthreads();

        // logS += CVOL @ Z  (K = 64 in two 32-deep MFMA chunks)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
            const int k0 = ks * 32 + fq * 8;
#pragma unroll
            for (int mt = 0; mt < 4; ++mt) {
                const int arow = mt * 16 + fr;
                mc_bf16x8 a = *reinterpret_cast<const mc_bf16x8*>(
                    &lds_m[arow * MCM_KP + k0]);
#pragma unroll
                for (int ct = 0; ct < 2; ++ct) {
                    const int bp = w * 32 + ct * 16 + fr;
                    mc_bf16x8 b = *reinterpret_cast<const mc_bf16x8*>(
                        &lds_zt[bp * MCM_KP + k0]);
                    logS[mt][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, b, logS[mt][ct], 0, 0, 0);
                }
            }
        }
        __syncthreads();   // Z^T reused next step

        // drift + portfolio value + drawdown, per fragment column
#pragma unroll
        for (int ct = 0; ct < 2; ++ct) {
            float part = 0.0f;
#pragma unroll
            for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    logS[mt][ct][r] += r_drift[mt][r];
                    part += r_w[mt][r] * __builtin_amdgcn_exp2f(logS[mt][ct][r]);
                }
            }
            // sum over the 4 lanes sharing this column (l ^ 16, l ^ 32)
            part += __shfl_xor(part, 16, 64);
            part += __shfl_xor(part, 32, 64);
            V[ct] = part;
            vmax[ct] = fmaxf(vmax[ct], part);
            mdd[ct] = fmaxf(mdd[ct], (vmax[ct] - part) * __builtin_amdgcn_rcpf(vmax[ct]));
        }
    }

    if (fq == 0) {
#pragma unroll
        for (int ct = 0; ct < 2; ++ct) {
            const long p = path0 + w * 32 + ct * 16 + fr;
            if (p < n_paths) {
                final_value[p] = V[ct];
                max_dd[p] = mdd[ct];
            }
        }
    }
}

}  // namespace

extern "C" void launch_mc_paths_mfma(const float* cvol, const float* drift,
                                     const float* wS0, float* final_value,
                                     float* max_dd, int n_assets,
                                     int n_steps, long n_paths, float v0,
                                     uint64_t seed, long path_base,
                                     int antithetic, hipStream_t stream) {
    if (n_assets != MCM_A)
        throw std::runtime_error("mc_paths_mfma: n_assets must be 64");
    if (n_paths % MCM_PB != 0)
        throw std::runtime_error(
            "mc_paths_mfma: n_paths must be a multiple of 256");
    long blocks = n_paths / MCM_PB;
    hipLaunchKernelGGL(mc_paths_mfma_kernel, dim3((unsigned)blocks),
                       dim3(512), 0, stream, cvol, drift, wS0, final_value,
                       max_dd, n_steps, n_paths, v0, seed, path_base,
                       antithetic);
}

extern "C" void launch_mc_paths(const float* cvol, const float* drift,
                                const float* wS0, const float* weights_unused,
                                float* final_value, float* max_dd,
                                int n_assets, int n_steps, long n_paths,
                                float v0, uint64_t seed, long path_base,
                                int antithetic, hipStream_t stream) {
    (void)weights_unused;
    long want = (n_paths + 255) / 256;
    int blocks = (int)(want < 8192 ? want : 8192);
    dim3 grid(blocks), block(256);
#define MC_CASE(AA)                                                           \
    case AA:                                                                  \
        hipLaunchKernelGGL(mc_paths_kernel<AA>, grid, block, 0, stream, cvol, \
                           drift, wS0, final_value, max_dd, n_steps, n_paths, \
                           v0, seed, path_base, antithetic);                  \
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

// ---------------------------------------------------------------------------
// Historical bootstrap simulation (reference monte_carlo_service.py:275-298
// 'historical' method): paths resample observed log-return ROWS with
// replacement instead of drawing GBM normals. One lane = one portfolio
// path; the same sampled time index applies to every asset of a step, so
// the empirical cross-asset correlation is preserved exactly (the
// reference bootstraps per asset independently and then ignores
// correlation in its portfolio sum — joint row resampling is the
// correlation-correct version of the same estimator). The (T_hist, A)
// log-return table is read through L2 (well under a slice of it); the
// Philox counter is the (path, step) pair so streams are reproducible
// and DP-shardable exactly like the GBM kernel.
// ---------------------------------------------------------------------------

namespace {

template <int A>
__global__ void __launch_bounds__(256) mc_bootstrap_kernel(
    const float* __restrict__ logret,   // (T_hist, A) log2-returns
    const float* __restrict__ wS0,      // (A,) weight * S0
    float* __restrict__ final_value,    // (n_paths,)
    float* __restrict__ max_dd,
    int t_hist, int n_steps, long n_paths, float v0, uint64_t seed,
    long path_base)
{
    __shared__ float lds_w[A];
    for (int i = threadIdx.x; i < A; i += blockDim.x)
        lds_w[i] = wS0[i];
    __syncthreads();

    const long stride = (long)gridDim.x * blockDim.x;
    for (long path = (long)blockIdx.x * blockDim.x + threadIdx.x;
         path < n_paths; path += stride) {
        float logS[A];
#pragma unroll
        for (int a = 0; a < A; ++a) logS[a] = 0.0f;
        float vmax = v0, mdd = 0.0f, V = v0;

        for (int s4 = 0; s4 < (n_steps + 3) / 4; ++s4) {
            Philox4 r = philox4x32(seed, (uint64_t)(path + path_base),
                                   0x8000000000000000ull | (uint64_t)s4);
            const uint32_t ix[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int step = s4 * 4 + j;
                if (step >= n_steps) break;
                const int t = (int)(ix[j] % (uint32_t)t_hist);
                const float* row = logret + (long)t * A;
                V = 0.0f;
#pragma unroll
                for (int a = 0; a < A; ++a) {
                    logS[a] += row[a];
                    V += lds_w[a] * __builtin_amdgcn_exp2f(logS[a]);
                }
                vmax = fmaxf(vmax, V);
                mdd = fmaxf(mdd,
                            (vmax - V) * __builtin_amdgcn_rcpf(vmax));
            }
        }
        final_value[path] = V;
        max_dd[path] = mdd;
    }
}

}  // namespace

extern "C" void launch_mc_bootstrap(const float* logret, const float* wS0,
                                    float* final_value, float* max_dd,
                                    int n_assets, int t_hist, int n_steps,
                                    long n_paths, float v0, uint64_t seed,
                                    long path_base, hipStream_t stream) {
    long want = (n_paths + 255) / 256;
    int blocks = (int)(want < 8192 ? want : 8192);
    dim3 grid(blocks), block(256);
#define MCB_CASE(AA)                                                          \
    case AA:                                                                  \
        hipLaunchKernelGGL(mc_bootstrap_kernel<AA>, grid, block, 0, stream,   \
                           logret, wS0, final_value, max_dd, t_hist,          \
                           n_steps, n_paths, v0, seed, path_base);            \
        break;
    switch (n_assets) {
        MCB_CASE(4)
        MCB_CASE(8)
        MCB_CASE(16)
        MCB_CASE(32)
        MCB_CASE(64)
        default:
            throw std::runtime_error(
                "mc_bootstrap: n_assets must be one of 4/8/16/32/64");
    }
#undef MCB_CASE
}
