// Fused GRU sequence kernels — sibling of lstm.hip (same decomposition:
// batch-tile 64 per block, wave w owns hidden columns [16w,16w+16) across
// the 3 gates [r|z|n], W_hh LDS-resident, per-step gate GEMM on
// v_mfma_f32_16x16x32_bf16, h carried through LDS).
//
// PyTorch GRU semantics (models/gru.py reference):
//   hp = h @ W_hh + b_hh                      (3H)
//   r = sigmoid(xp_r + hp_r)   z = sigmoid(xp_z + hp_z)
//   n = tanh(xp_n + r * hp_n)
//   h' = (1 - z) * n + z * h
// Backward outputs the x-side pre-activation grads (da_r, da_z, da_n);
// the h-side differs only in the n column (da_n * r), applied on the
// torch side for the dW_hh GEMM (models/gru.py).

#include "common.hpp"

typedef __bf16 bf16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define GRU_MT 4            // M-tiles per wave (batch tile = 16*MT rows)
#define GRU_BM (16 * GRU_MT)

namespace {

// 1-ulp v_rcp instead of IEEE div (see lstm.hip rationale)
DEV_INLINE float gsigmoid(float x) {
    return __builtin_amdgcn_rcpf(1.0f + __expf(-x));
}
DEV_INLINE float gtanh(float x) {
    return 2.0f * __builtin_amdgcn_rcpf(1.0f + __expf(-2.0f * x)) - 1.0f;
}

template <int H>
__global__ void __launch_bounds__((H / 16) * 64) gru_seq_fwd_kernel(
    const bf16* __restrict__ xproj,    // (T, B, 3H)  x@W_ih + b_ih
    const bf16* __restrict__ Wt,       // (3H, H)     W_hh^T n-major
    const float* __restrict__ bias,    // (3H,)       b_hh
    bf16* __restrict__ h_out,          // (T, B, H)
    bf16* __restrict__ gates_out,      // (T, B, 3H)  post-act r, z, n
    float* __restrict__ hpn_out,       // (T, B, H)   hp_n (pre-r-multiply)
    int B, int T, int save_mode)       // 0 = inference: skip bwd saves
{
    constexpr int NW = H / 16;
    constexpr int THREEH = 3 * H;
    constexpr int HP = H + 8;

    __shared__ bf16 lds_h[GRU_BM * HP];
    __shared__ bf16 lds_w[THREEH * HP];

    const int b0 = blockIdx.x * GRU_BM;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;
    const int fr = lane & 15;
    const int fq = lane >> 4;

    for (int i = tid; i < THREEH * H; i += NW * 64)
        lds_w[(i / H) * HP + (i % H)] = Wt[i];
    for (int i = tid; i < GRU_BM * HP; i += NW * 64)
        lds_h[i] = (bf16)0.0f;
    __syncthreads();

    const int ncol = 16 * w + fr;
    float bias_g[3];
#pragma unroll
    for (int g = 0; g < 3; ++g) bias_g[g] = bias[g * H + ncol];

    // previous h at this lane's fragment positions (for h' = ... + z*h)
    float hprev[GRU_MT][4];
#pragma unroll
    for (int mt = 0; mt < GRU_MT; ++mt)
#pragma unroll
        for (int r = 0; r < 4; ++r) hprev[mt][r] = 0.0f;

    for (int t = 0; t < T; ++t) {
        const long base_tb = ((long)t * B + b0);
        f32x4 acc[GRU_MT][3];            // hp = h @ W_hh + b_hh
#pragma unroll
        for (int mt = 0; mt < GRU_MT; ++mt)
#pragma unroll
            for (int g = 0; g < 3; ++g) {
#pragma unroll
                for (int r = 0; r < 4; ++r) acc[mt][g][r] = bias_g[g];
            }
#pragma unroll
        for (int ks = 0; ks < H / 32; ++ks) {
            const int k0 = ks * 32 + fq * 8;
#pragma unroll
            for (int mt = 0; mt < GRU_MT; ++mt) {
                const int arow = mt * 16 + fr;
                bf16x8 a = *reinterpret_cast<const bf16x8*>(
                    &lds_h[arow * HP + k0]);
#pragma unroll
                for (int g = 0; g < 3; ++g) {
                    const int bcol = g * H + 16 * w + fr;
                    bf16x8 b = *reinterpret_cast<const bf16x8*>(
                        &lds_w[bcol * HP + k0]);
                    acc[mt][g] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, b, acc[mt][g], 0, 0, 0);
                }
            }
        }
        __syncthreads();

#pragma unroll
        for (int mt = 0; mt < GRU_MT; ++mt) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = mt * 16 + fq * 4 + r;
                const long xb = (base_tb + row) * THREEH;
                const float xr = (float)xproj[xb + 0 * H + ncol];
                const float xz = (float)xproj[xb + 1 * H + ncol];
                const float xn = (float)xproj[xb + 2 * H + ncol];
                const float gr = gsigmoid(xr + acc[mt][0][r]);
                const float gz = gsigmoid(xz + acc[mt][1][r]);
                const float hpn = acc[mt][2][r];
                const float gn = gtanh(xn + gr * hpn);
                const float h =
                    (1.0f - gz) * gn + gz * hprev[mt][r];
                hprev[mt][r] = h;
                const bf16 hb = (bf16)h;
                lds_h[row * HP + ncol] = hb;
                if (save_mode) {       // training: save for backward
                    const long gb = (base_tb + row) * THREEH;
                    gates_out[gb + 0 * H + ncol] = (bf16)gr;
                    gates_out[gb + 1 * H + ncol] = (bf16)gz;
                    gates_out[gb + 2 * H + ncol] = (bf16)gn;
                    hpn_out[(base_tb + row) * H + ncol] = hpn;
                }
                h_out[(base_tb + row) * H + ncol] = hb;
            }
        }
        __syncthreads();
    }
}

template <int H>
__global__ void __launch_bounds__((H / 16) * 64) gru_seq_bwd_kernel(
    const float* __restrict__ dh_up,   // (T, B, H)
    const bf16* __restrict__ gates,    // (T, B, 3H) saved r, z, n
    const float* __restrict__ hpn_sav, // (T, B, H)
    const bf16* __restrict__ h_out,    // (T, B, H)  (h_prev = shifted)
    const bf16* __restrict__ W,        // (H, 3H) row-major
    bf16* __restrict__ dgates_out,     // (T, B, 3H) x-side grads
    int B, int T)
{
    constexpr int NW = H / 16;
    constexpr int THREEH = 3 * H;
    constexpr int GP = THREEH + 8;

    __shared__ bf16 lds_dg[GRU_BM * GP];   // h-side dgates (GEMM A operand)
    __shared__ bf16 lds_w[H * GP];

    const int b0 = blockIdx.x * GRU_BM;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;
    const int fr = lane & 15;
    const int fq = lane >> 4;

    for (int i = tid; i < H * THREEH; i += NW * 64)
        lds_w[(i / THREEH) * GP + (i % THREEH)] = W[i];
    __syncthreads();

    const int ncol = 16 * w + fr;

    float dhrec[GRU_MT][4];
#pragma unroll
    for (int mt = 0; mt < GRU_MT; ++mt)
#pragma unroll
        for (int r = 0; r < 4; ++r) dhrec[mt][r] = 0.0f;

    for (int t = T - 1; t >= 0; --t) {
        const long base_tb = ((long)t * B + b0);
#pragma unroll
        for (int mt = 0; mt < GRU_MT; ++mt) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = mt * 16 + fq * 4 + r;
                const long gb = (base_tb + row) * THREEH;
                const float gr = (float)gates[gb + 0 * H + ncol];
                const float gz = (float)gates[gb + 1 * H + ncol];
                const float gn = (float)gates[gb + 2 * H + ncol];
                const float hpn = hpn_sav[(base_tb + row) * H + ncol];
                const float hprev =
                    (t > 0) ? (float)h_out[((base_tb - B) + row) * H + ncol]
                            : 0.0f;
                const float dh =
                    dh_up[(base_tb + row) * H + ncol] + dhrec[mt][r];
                const float dz = dh * (hprev - gn);
                const float dn = dh * (1.0f - gz);
                float dh_direct = dh * gz;           // h' = ... + z*h term
                const float da_n = dn * (1.0f - gn * gn);
                const float dr = da_n * hpn;
                const float da_r = dr * gr * (1.0f - gr);
                const float da_z = dz * gz * (1.0f - gz);
                const float dhp_n = da_n * gr;       // h-side n-column
                const bf16 b_r = (bf16)da_r, b_z = (bf16)da_z;
                const bf16 b_nx = (bf16)da_n, b_nh = (bf16)dhp_n;
                lds_dg[row * GP + 0 * H + ncol] = b_r;
                lds_dg[row * GP + 1 * H + ncol] = b_z;
                lds_dg[row * GP + 2 * H + ncol] = b_nh;
                dgates_out[gb + 0 * H + ncol] = b_r;
                dgates_out[gb + 1 * H + ncol] = b_z;
                dgates_out[gb + 2 * H + ncol] = b_nx;
                dhrec[mt][r] = dh_direct;     // + GEMM contribution below
            }
        }
        __syncthreads();

        // dh_prev += dgates_h @ W_hh^T : (64 x 3H) @ (3H x H)
        f32x4 acc[GRU_MT];
#pragma unroll
        for (int mt = 0; mt < GRU_MT; ++mt) acc[mt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll 3
        for (int ks = 0; ks < THREEH / 32; ++ks) {
            const int k0 = ks * 32 + fq * 8;
#pragma unroll
            for (int mt = 0; mt < GRU_MT; ++mt) {
                const int arow = mt * 16 + fr;
                bf16x8 a = *reinterpret_cast<const bf16x8*>(
                    &lds_dg[arow * GP + k0]);
                bf16x8 b = *reinterpret_cast<const bf16x8*>(
                    &lds_w[ncol * GP + k0]);
                acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a, b, acc[mt], 0, 0, 0);
            }
        }
#pragma unroll
        for (int mt = 0; mt < GRU_MT; ++mt)
#pragma unroll
            for (int r = 0; r < 4; ++r) dhrec[mt][r] += acc[mt][r];
        __syncthreads();
    }
}

}  // namespace

extern "C" void launch_gru_seq_fwd(const void* xproj, const void* Wt,
                                   const float* bias, void* h_out,
                                   void* gates_out, float* hpn_out, int B,
                                   int T, int H, int save_mode,
                                   hipStream_t stream) {
    if (B % GRU_BM != 0)
        throw std::runtime_error("gru_fwd: B must be a multiple of 64");
    dim3 grid(B / GRU_BM);
    if (H == 64) {
        hipLaunchKernelGGL(gru_seq_fwd_kernel<64>, grid, dim3(256), 0,
                           stream, (const bf16*)xproj, (const bf16*)Wt,
                           bias, (bf16*)h_out, (bf16*)gates_out, hpn_out,
                           B, T, save_mode);
    } else if (H == 32) {
        hipLaunchKernelGGL(gru_seq_fwd_kernel<32>, grid, dim3(128), 0,
                           stream, (const bf16*)xproj, (const bf16*)Wt,
                           bias, (bf16*)h_out, (bf16*)gates_out, hpn_out,
                           B, T, save_mode);
    } else {
        throw std::runtime_error("gru_fwd: H must be 32 or 64");
    }
}

extern "C" void launch_gru_seq_bwd(const float* dh_up, const void* gates,
                                   const float* hpn_sav, const void* h_out,
                                   const void* W, void* dgates_out, int B,
                                   int T, int H, hipStream_t stream) {
    if (B % GRU_BM != 0)
        throw std::runtime_error("gru_bwd: B must be a multiple of 64");
    dim3 grid(B / GRU_BM);
    if (H == 64) {
        hipLaunchKernelGGL(gru_seq_bwd_kernel<64>, grid, dim3(256), 0,
                           stream, dh_up, (const bf16*)gates, hpn_sav,
                           (const bf16*)h_out, (const bf16*)W,
                           (bf16*)dgates_out, B, T);
    } else if (H == 32) {
        hipLaunchKernelGGL(gru_seq_bwd_kernel<32>, grid, dim3(128), 0,
                           stream, dh_up, (const bf16*)gates, hpn_sav,
                           (const bf16*)h_out, (const bf16*)W,
                           (bf16*)dgates_out, B, T);
    } else {
        throw std::runtime_error("gru_bwd: H must be 32 or 64");
    }
}
