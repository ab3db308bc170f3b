// Portfolio covariance via f32 MFMA.
//
// Replaces portfolio_risk_service.py:286-326 (`returns_df.corr()` on
// pandas) with an MFMA GEMM: Sxy = X^T X accumulated with
// v_mfma_f32_16x16x4_f32 (exact f32 at the f32 vector rate — no xf32 on
// gfx950), then cov[i][j] = (Sxy - Sx_i Sx_j / T) / (T-1) in a finalize
// kernel. The quadratic-form portfolio VaR (sqrt(w^T (vv^T o C) w),
// portfolio_risk_service.py:328-398) runs on these outputs in
// ops/covar.py.
//
// Fragment mapping for mfma_f32_16x16x4f32 (cdna_hip_programming.md §3):
//   A[i][k]: lane l holds A[l&15][l>>4]   (one f32 VGPR)
//   B[k][j]: lane l holds B[l>>4][l&15]
//   C/D:     col = lane&15, row = (lane>>4)*4 + reg,  reg in [0,4)
// mfma_gemm_test_kernel below validates this mapping on-device against
// torch.matmul (tests/test_gpu_kernels.py).

#include "common.hpp"

using f32x4 = __attribute__((ext_vector_type(4))) float;

#define COV_TILE_T 64
#define COV_PAD 80   // LDS row stride (16-lane groups land on disjoint banks)

namespace {

__global__ void __launch_bounds__(256) cov_accum_kernel(
    const float* __restrict__ X,   // (T, N) row-major returns
    float* __restrict__ Sxy,       // (N, N) f32, pre-zeroed, atomic accum
    float* __restrict__ Sx,        // (N,)   f32, pre-zeroed
    int T, int N)
{
    __shared__ float tile[COV_TILE_T * COV_PAD];
    __shared__ float lds_sx[64];

    const int wid = threadIdx.x / WAVE;        // 4 waves
    const int lane = threadIdx.x & (WAVE - 1);
    const int ncol_tiles = N / 16;

    // wave w owns output rows [16w, 16w+16) x all column tiles
    f32x4 acc[4];                               // up to N=64 -> 4 col tiles
    for (int c = 0; c < 4; ++c) acc[c] = {0.f, 0.f, 0.f, 0.f};
    float colsum = 0.0f;
    const int my_col = threadIdx.x % N;
    const int my_row0 = threadIdx.x / N;        // 256/N rows per pass
    const int rows_per_pass = 256 / N;

    if (threadIdx.x < 64) lds_sx[threadIdx.x] = 0.0f;

    const int ntiles = (T + COV_TILE_T - 1) / COV_TILE_T;
    for (int tile_i = blockIdx.x; tile_i < ntiles; tile_i += gridDim.x) {
        const int t0 = tile_i * COV_TILE_T;
        const int nrow = min(COV_TILE_T, T - t0);
        __syncthreads();
        // stage: rows t0..t0+nrow of X into LDS (padded stride)
        for (int i = threadIdx.x; i < nrow * N; i += blockDim.x) {
            int r = i / N, c = i % N;
            tile[r * COV_PAD + c] = X[(long)(t0 + r) * N + c];
        }
        if (nrow < COV_TILE_T) {   // zero-pad the K tail
            for (int i = threadIdx.x; i < (COV_TILE_T - nrow) * N;
                 i += blockDim.x) {
                int r = nrow + i / N, c = i % N;
                tile[r * COV_PAD + c] = 0.0f;
            }
        }
        __syncthreads();

        const int arow = 16 * wid + (lane & 15);
        if (arow < N) {
#pragma unroll 4
            for (int k0 = 0; k0 < COV_TILE_T; k0 += 4) {
                const int kr = k0 + (lane >> 4);
                const float a = tile[kr * COV_PAD + arow];
                // c must stay a compile-time index: runtime-indexed
                // ext_vector arrays spill to scratch (guide §5.4 rule 20)
#pragma unroll
                for (int c = 0; c < 4; ++c) {
                    if (c >= ncol_tiles) break;
                    const float b = tile[kr * COV_PAD + 16 * c + (lane & 15)];
                    acc[c] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b,
                                                                  acc[c], 0,
                                                                  0, 0);
                }
            }
        }
        // column sums (for the mean term)
        for (int r = my_row0; r < nrow; r += rows_per_pass)
            colsum += tile[r * COV_PAD + my_col];
    }

    __syncthreads();
    atomicAdd(&lds_sx[my_col], colsum);
    __syncthreads();
    if (threadIdx.x < N) atomicAdd(&Sx[threadIdx.x], lds_sx[threadIdx.x]);

    const int orow_base = 16 * wid + (lane >> 4) * 4;
    if (orow_base < N) {
#pragma unroll
        for (int c = 0; c < 4; ++c) {
            if (c >= ncol_tiles) break;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                atomicAdd(&Sxy[(long)(orow_base + r) * N + 16 * c +
                               (lane & 15)],
                          acc[c][r]);
            }
        }
    }
}

__global__ void cov_finalize_kernel(float* __restrict__ Sxy,
                                    const float* __restrict__ Sx, int T,
                                    int N) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= N * N) return;
    int r = i / N, c = i % N;
    float t = (float)T;
    float v = (Sxy[i] - Sx[r] * Sx[c] / t) / (t - 1.0f);
    Sxy[i] = v;
}

// --- layout validation GEMM: C(M,N) = A(M,K) @ B(K,N), all f32 ----------
__global__ void mfma_gemm_test_kernel(const float* __restrict__ A,
                                      const float* __restrict__ B,
                                      float* __restrict__ C, int M, int N,
                                      int K) {
    const int i0 = blockIdx.x * 16, j0 = blockIdx.y * 16;
    const int lane = threadIdx.x & (WAVE - 1);
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = 0; k0 < K; k0 += 4) {
        float a = A[(long)(i0 + (lane & 15)) * K + k0 + (lane >> 4)];
        float b = B[(long)(k0 + (lane >> 4)) * N + j0 + (lane & 15)];
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r)
        C[(long)(i0 + (lane >> 4) * 4 + r) * N + j0 + (lane & 15)] = acc[r];
}

}  // namespace

extern "C" void launch_cov(const float* X, float* cov, int T, int N,
                           hipStream_t stream) {
    if (N > 64 || (N & 15) != 0)
        throw std::runtime_error("cov: N must be a multiple of 16, <= 64");
    float* Sx = nullptr;
    if (hipMallocAsync((void**)&Sx, N * sizeof(float), stream) != hipSuccess)
        throw std::runtime_error("cov: workspace alloc failed");
    if (hipMemsetAsync(cov, 0, (size_t)N * N * sizeof(float), stream)
            != hipSuccess ||
        hipMemsetAsync(Sx, 0, N * sizeof(float), stream) != hipSuccess)
        throw std::runtime_error("cov: workspace memset failed");
    int ntiles = (T + COV_TILE_T - 1) / COV_TILE_T;
    int grid = ntiles < 512 ? ntiles : 512;
    hipLaunchKernelGGL(cov_accum_kernel, dim3(grid), dim3(256), 0, stream, X,
                       cov, Sx, T, N);
    hipLaunchKernelGGL(cov_finalize_kernel, dim3((N * N + 255) / 256),
                       dim3(256), 0, stream, cov, Sx, T, N);
    (void)hipFreeAsync(Sx, stream);
}

extern "C" void launch_mfma_gemm_test(const void* A, const void* B, float* C,
                                      int M, int N, int K,
                                      hipStream_t stream) {
    dim3 grid(M / 16, N / 16);
    hipLaunchKernelGGL(mfma_gemm_test_kernel, grid, dim3(64), 0, stream,
                       (const float*)A, (const float*)B, C, M, N, K);
}
