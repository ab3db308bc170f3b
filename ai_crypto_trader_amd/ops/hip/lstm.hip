// Fused LSTM sequence kernels (BASELINE config #2).
//
// Replaces neural_network_service.py's Keras LSTM stack (:164-421) with a
// hand-written gfx950 recurrent cell: ONE kernel launch marches the whole
// sequence, W_hh stays resident in LDS, h feeds back through LDS, and the
// per-step gate GEMM (batch-tile 64 x 4H x H) runs on
// v_mfma_f32_16x16x32_bf16 with the input projection (x @ W_ih, a plain
// hipBLASLt GEMM on the torch side) pre-computed and read as the
// accumulator init. No Triton, no CUDA shims, no per-step launches.
//
// Decomposition (template H, NW = H/16 waves per block):
//   block = 64-row batch tile; wave w owns hidden columns [16w, 16w+16)
//   ACROSS all four gates, so a lane holds i,f,g,o for the same (row, n)
//   and the cell update c = f*c + i*g, h = o*tanh(c) is register-local.
//   c lives in registers across the whole sequence; h round-trips LDS
//   (padded stride -> conflict-free b128 fragment reads).
//
// Fragment mapping mfma_f32_16x16x32_bf16 (validated on-device by
// mfma_gemm_test_bf16 vs torch.matmul):
//   A[m][k]: lane l holds A[l&15][8*(l>>4) + j], j=0..7  (one b128 read)
//   B[k][n]: lane l holds B[8*(l>>4)+j][l&15]  -> store B n-major so the
//            8 k-elements are contiguous per lane
//   C[m][n]: lane l, reg r -> m = (l>>4)*4 + r, n = l&15
//
// Backward: reverse-time kernel computes the sequential part only
// (per-step dgates + the dh_rec = dgates @ W_hh^T chain). The big
// batched reductions dW_hh = sum_t h_{t-1}^T dgates_t, dW_ih, db are
// single hipBLASLt GEMMs on the torch side (models/lstm.py).

#include "common.hpp"

typedef __bf16 bf16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define LSTM_MT 4           // M-tiles per wave (batch tile = 16*MT rows)
#define LSTM_BM (16 * LSTM_MT)   // 32-row batch tiles: 2x the blocks of a
                                 // 64-row tile -> 2 blocks/CU at B=16384
                                 // (B/64 blocks was exactly 1/CU: latency-bound)

namespace {

// 1-ulp v_rcp instead of IEEE div: the gate math is tolerance-tested
// against PyTorch fp32 (not bitwise), and libm-grade division costs
// ~10 instrs per gate element per timestep.
DEV_INLINE float sigmoidf_(float x) {
    return __builtin_amdgcn_rcpf(1.0f + __expf(-x));
}
DEV_INLINE float tanhf_(float x) {
    // tanh(x) = 2*sigmoid(2x) - 1, rcp as above
    return 2.0f * __builtin_amdgcn_rcpf(1.0f + __expf(-2.0f * x)) - 1.0f;
}

template <int H>
__global__ void __launch_bounds__((H / 16) * 64) lstm_seq_fwd_kernel(
    const bf16* __restrict__ xproj,    // (T, B, 4H)  x@W_ih + b_ih (bf16)
    const bf16* __restrict__ Wt,       // (4H, H)     W_hh transposed, n-major
    const float* __restrict__ bias,    // (4H,)       b_hh
    bf16* __restrict__ h_out,          // (T, B, H)
    bf16* __restrict__ gates_out,      // (T, B, 4H)  post-activation i,f,g,o
    float* __restrict__ c_out,         // (T, B, H)
    int B, int T, int save_mode)       // 0: inference (h only), 1: training
{
    constexpr int NW = H / 16;          // waves per block
    constexpr int FOURH = 4 * H;
    constexpr int HP = H + 8;           // padded LDS strides (bank-spread)
    constexpr int KST = H / 32;         // MFMA K-steps per gate tile

    __shared__ bf16 lds_h[LSTM_BM * HP];
    __shared__ bf16 lds_w[FOURH * HP];

    const int b0 = blockIdx.x * LSTM_BM;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;             // wave id = hidden sub-column
    const int fr = lane & 15;           // fragment col
    const int fq = lane >> 4;           // fragment row group / k group

    // stage W_hh^T (4H x H) into LDS, padded rows
    for (int i = tid; i < FOURH * H; i += NW * 64)
        lds_w[(i / H) * HP + (i % H)] = Wt[i];
    // zero-init h
    for (int i = tid; i < LSTM_BM * HP; i += NW * 64)
        lds_h[i] = (bf16)0.0f;
    __syncthreads();

    const int ncol = 16 * w + fr;       // this lane's hidden column
    float bias_g[4];
#pragma unroll
    for (int g = 0; g < 4; ++g) bias_g[g] = bias[g * H + ncol];

    float c_reg[LSTM_MT][4];            // [m_tile][reg] cell state
#pragma unroll
    for (int mt = 0; mt < LSTM_MT; ++mt)
#pragma unroll
        for (int r = 0; r < 4; ++r) c_reg[mt][r] = 0.0f;

    for (int t = 0; t < T; ++t) {
        // ---- gates = h @ W_hh^T(+) xproj + bias --------------------------
        f32x4 acc[LSTM_MT][4];          // [m_tile][gate]
        const long base_tb = ((long)t * B + b0);
#pragma unroll
        for (int mt = 0; mt < LSTM_MT; ++mt) {
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                const int col = g * H + ncol;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int row = mt * 16 + fq * 4 + r;
                    acc[mt][g][r] =
                        (float)xproj[(base_tb + row) * FOURH + col] +
                        bias_g[g];
                }
            }
        }
#pragma unroll
        for (int ks = 0; ks < KST; ++ks) {
            const int k0 = ks * 32 + fq * 8;
#pragma unroll
            for (int mt = 0; mt < LSTM_MT; ++mt) {
                const int arow = mt * 16 + fr;
                bf16x8 a = *reinterpret_cast<const bf16x8*>(
                    &lds_h[arow * HP + k0]);
#pragma unroll
                for (int g = 0; g < 4; ++g) {
                    const int bcol = g * H + 16 * w + fr;
                    bf16x8 b = *reinterpret_cast<const bf16x8*>(
                        &lds_w[bcol * HP + k0]);
                    acc[mt][g] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, b, acc[mt][g], 0, 0, 0);
                }
            }
        }
        __syncthreads();   // all reads of lds_h done before overwrite

        // ---- cell update + write h ---------------------------------------
#pragma unroll
        for (int mt = 0; mt < LSTM_MT; ++mt) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = mt * 16 + fq * 4 + r;
                const float gi = sigmoidf_(acc[mt][0][r]);
                const float gf = sigmoidf_(acc[mt][1][r]);
                const float gg = tanhf_(acc[mt][2][r]);
                const float go = sigmoidf_(acc[mt][3][r]);
                const float c = gf * c_reg[mt][r] + gi * gg;
                c_reg[mt][r] = c;
                const float h = go * tanhf_(c);
                const bf16 hb = (bf16)h;
                lds_h[row * HP + ncol] = hb;
                if (save_mode) {       // backward saves: ~5x the stores of
                                       // the inference path, skip them there
                    const long gb = (base_tb + row) * FOURH;
                    gates_out[gb + 0 * H + ncol] = (bf16)gi;
                    gates_out[gb + 1 * H + ncol] = (bf16)gf;
                    gates_out[gb + 2 * H + ncol] = (bf16)gg;
                    gates_out[gb + 3 * H + ncol] = (bf16)go;
                    c_out[(base_tb + row) * H + ncol] = c;
                }
                h_out[(base_tb + row) * H + ncol] = hb;
            }
        }
        __syncthreads();
    }
}

template <int H>
__global__ void __launch_bounds__((H / 16) * 64) lstm_seq_bwd_kernel(
    const float* __restrict__ dh_up,   // (T, B, H) upstream grad
    const bf16* __restrict__ gates,    // (T, B, 4H) saved i,f,g,o
    const float* __restrict__ c_sav,   // (T, B, H)
    const bf16* __restrict__ W,        // (H, 4H)  W_hh row-major
    bf16* __restrict__ dgates_out,     // (T, B, 4H)
    int B, int T)
{
    constexpr int NW = H / 16;
    constexpr int FOURH = 4 * H;
    constexpr int HP = H + 8;
    constexpr int GP = FOURH + 8;

    __shared__ bf16 lds_dg[LSTM_BM * GP];   // dgates tile (A operand)
    __shared__ bf16 lds_w[H * GP];          // W_hh (B operand: k=4H contig)

    const int b0 = blockIdx.x * LSTM_BM;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;
    const int fr = lane & 15;
    const int fq = lane >> 4;

    for (int i = tid; i < H * FOURH; i += NW * 64)
        lds_w[(i / FOURH) * GP + (i % FOURH)] = W[i];
    __syncthreads();

    const int ncol = 16 * w + fr;

    float dc[LSTM_MT][4];
    float dhrec[LSTM_MT][4];
#pragma unroll
    for (int mt = 0; mt < LSTM_MT; ++mt)
#pragma unroll
        for (int r = 0; r < 4; ++r) { dc[mt][r] = 0.0f; dhrec[mt][r] = 0.0f; }

    for (int t = T - 1; t >= 0; --t) {
        const long base_tb = ((long)t * B + b0);
#pragma unroll
        for (int mt = 0; mt < LSTM_MT; ++mt) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = mt * 16 + fq * 4 + r;
                const long gb = (base_tb + row) * FOURH;
                const float gi = (float)gates[gb + 0 * H + ncol];
                const float gf = (float)gates[gb + 1 * H + ncol];
                const float gg = (float)gates[gb + 2 * H + ncol];
                const float go = (float)gates[gb + 3 * H + ncol];
                const float c = c_sav[(base_tb + row) * H + ncol];
                const float cprev =
                    (t > 0) ? c_sav[((base_tb - B) + row) * H + ncol] : 0.0f;
                const float tc = tanhf_(c);
                const float dh =
                    dh_up[(base_tb + row) * H + ncol] + dhrec[mt][r];
                const float do_ = dh * tc;
                float dcv = dc[mt][r] + dh * go * (1.0f - tc * tc);
                const float di = dcv * gg;
                const float df = dcv * cprev;
                const float dgg = dcv * gi;
                dc[mt][r] = dcv * gf;               // dc_{t-1}
                const float dai = di * gi * (1.0f - gi);
                const float daf = df * gf * (1.0f - gf);
                const float dag = dgg * (1.0f - gg * gg);
                const float dao = do_ * go * (1.0f - go);
                const bf16 b_i = (bf16)dai, b_f = (bf16)daf;
                const bf16 b_g = (bf16)dag, b_o = (bf16)dao;
                lds_dg[row * GP + 0 * H + ncol] = b_i;
                lds_dg[row * GP + 1 * H + ncol] = b_f;
                lds_dg[row * GP + 2 * H + ncol] = b_g;
                lds_dg[row * GP + 3 * H + ncol] = b_o;
                dgates_out[gb + 0 * H + ncol] = b_i;
                dgates_out[gb + 1 * H + ncol] = b_f;
                dgates_out[gb + 2 * H + ncol] = b_g;
                dgates_out[gb + 3 * H + ncol] = b_o;
            }
        }
        __syncthreads();

        // dh_rec = dgates @ W_hh^T : (BM x 4H) @ (4H x H) -> (BM x H)
        f32x4 acc[LSTM_MT];
#pragma unroll
        for (int mt = 0; mt < LSTM_MT; ++mt) acc[mt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll 4
        for (int ks = 0; ks < FOURH / 32; ++ks) {
            const int k0 = ks * 32 + fq * 8;
#pragma unroll
            for (int mt = 0; mt < LSTM_MT; ++mt) {
                const int arow = mt * 16 + fr;
                bf16x8 a = *reinterpret_cast<const bf16x8*>(
                    &lds_dg[arow * GP + k0]);
                // B[k][n] = W_hh[n][k]: row n of W, 8 contiguous k
                bf16x8 b = *reinterpret_cast<const bf16x8*>(
                    &lds_w[ncol * GP + k0]);
                acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a, b, acc[mt], 0, 0, 0);
            }
        }
#pragma unroll
        for (int mt = 0; mt < LSTM_MT; ++mt)
#pragma unroll
            for (int r = 0; r < 4; ++r) dhrec[mt][r] = acc[mt][r];
        __syncthreads();
    }
}

// --- bf16 MFMA layout validation GEMM: C = A @ B, f32 out ----------------
__global__ void mfma_gemm_test_bf16_kernel(const bf16* __restrict__ A,
                                           const bf16* __restrict__ B,
                                           float* __restrict__ C, int M,
                                           int N, int K) {
    const int i0 = blockIdx.x * 16, j0 = blockIdx.y * 16;
    const int lane = threadIdx.x & 63;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = 0; k0 < K; k0 += 32) {
        bf16x8 a, b;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            a[j] = A[(long)(i0 + (lane & 15)) * K + k0 + 8 * (lane >> 4) + j];
            b[j] = B[(long)(k0 + 8 * (lane >> 4) + j) * N + j0 + (lane & 15)];
        }
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r)
        C[(long)(i0 + (lane >> 4) * 4 + r) * N + j0 + (lane & 15)] = acc[r];
}

}  // namespace

extern "C" void launch_lstm_seq_fwd(const void* xproj, const void* Wt,
                                    const float* bias, void* h_out,
                                    void* gates_out, float* c_out, int B,
                                    int T, int H, int save_mode,
                                    hipStream_t stream) {
    if (B % LSTM_BM != 0)
        throw std::runtime_error("lstm_fwd: B must be a multiple of 64");
    dim3 grid(B / LSTM_BM);
    if (H == 64) {
        hipLaunchKernelGGL(lstm_seq_fwd_kernel<64>, grid, dim3(256), 0,
                           stream, (const bf16*)xproj, (const bf16*)Wt, bias,
                           (bf16*)h_out, (bf16*)gates_out, c_out, B, T,
                           save_mode);
    } else if (H == 32) {
        hipLaunchKernelGGL(lstm_seq_fwd_kernel<32>, grid, dim3(128), 0,
                           stream, (const bf16*)xproj, (const bf16*)Wt, bias,
                           (bf16*)h_out, (bf16*)gates_out, c_out, B, T,
                           save_mode);
    } else {
        throw std::runtime_error("lstm_fwd: H must be 32 or 64");
    }
}

extern "C" void launch_lstm_seq_bwd(const float* dh_up, const void* gates,
                                    const float* c_sav, const void* W,
                                    void* dgates_out, int B, int T, int H,
                                    hipStream_t stream) {
    if (B % LSTM_BM != 0)
        throw std::runtime_error("lstm_bwd: B must be a multiple of 64");
    dim3 grid(B / LSTM_BM);
    if (H == 64) {
        hipLaunchKernelGGL(lstm_seq_bwd_kernel<64>, grid, dim3(256), 0,
                           stream, dh_up, (const bf16*)gates, c_sav,
                           (const bf16*)W, (bf16*)dgates_out, B, T);
    } else if (H == 32) {
        hipLaunchKernelGGL(lstm_seq_bwd_kernel<32>, grid, dim3(128), 0,
                           stream, dh_up, (const bf16*)gates, c_sav,
                           (const bf16*)W, (bf16*)dgates_out, B, T);
    } else {
        throw std::runtime_error("lstm_bwd: H must be 32 or 64");
    }
}

extern "C" void launch_mfma_gemm_test_bf16(const void* A, const void* B,
                                           float* C, int M, int N, int K,
                                           hipStream_t stream) {
    dim3 grid(M / 16, N / 16);
    hipLaunchKernelGGL(mfma_gemm_test_bf16_kernel, grid, dim3(64), 0, stream,
                       (const bf16*)A, (const bf16*)B, C, M, N, K);
}
