// Fused multi-head attention forward for short sequences (S <= 64) —
// the attention/transformer price predictors' shape (seq_len 60, d_head
// 16-64; neural_network_service.py:236-306).
//
// One block = one (batch, head): Q/K/V tiles live entirely in LDS, scores
// S = scale * Q@K^T run on v_mfma_f32_16x16x32_bf16, the row softmax is a
// 16-lane shfl-xor reduction on the score fragments (no LDS round-trip for
// max/sum), P goes back through LDS as the PV GEMM's A operand, and V is
// staged TRANSPOSED so PV's B fragments are contiguous b128 reads. Whole
// sequences this short L2/LDS-fit, so flash-style tiling would be pure
// overhead (cdna_hip_programming.md common-mistake #7) — one tile IS the
// sequence.
//
// Backward: attn_bwd_kernel below — fully fused on the same fragments
// (dV = P^T dO, softmax-jacobian, dQ = dS K, dK = dS^T Q) consuming the
// softmax matrix P the forward saves when training.
//
// Layouts: Q,K,V,O (BH, S, D) row-major bf16 in global; S padded to 64
// rows with masked (-inf score) columns beyond s_len.

#include "common.hpp"

typedef __bf16 at_bf16;
typedef __bf16 at_bf16x8 __attribute__((ext_vector_type(8)));
typedef float at_f32x4 __attribute__((ext_vector_type(4)));

#define ATT_S 64            // padded sequence tile

namespace {

template <int D>
__global__ void __launch_bounds__(256) attn_fwd_kernel(
    const at_bf16* __restrict__ Q,   // (BH, S, D)
    const at_bf16* __restrict__ K,
    const at_bf16* __restrict__ V,
    at_bf16* __restrict__ O,         // (BH, S, D)
    at_bf16* __restrict__ P_out,     // (BH, 64, 64) or nullptr (training)
    int s_len, float scale, long bh_count)
{
    constexpr int KP = (D < 32 ? 32 : D);      // padded K-dim for MFMA
    constexpr int DP = KP + 8;                 // LDS row stride (bf16)
    constexpr int SP = ATT_S + 8;

    __shared__ at_bf16 lds_q[ATT_S * DP];
    __shared__ at_bf16 lds_k[ATT_S * DP];
    __shared__ at_bf16 lds_vt[KP * SP];        // V transposed: [d][s]
    __shared__ at_bf16 lds_p[ATT_S * SP];      // softmaxed scores

    const long bh = blockIdx.x;
    if (bh >= bh_count) return;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;                    // 4 waves: rows [16w,16w+16)
    const int fr = lane & 15;
    const int fq = lane >> 4;

    // ---- stage Q, K (zero-padded), V^T ---------------------------------
    const long base = bh * (long)s_len * D;
    for (int i = tid; i < ATT_S * KP; i += 256) {
        const int s = i / KP, d = i % KP;
        at_bf16 qv = (at_bf16)0.0f, kv = (at_bf16)0.0f, vv = (at_bf16)0.0f;
        if (s < s_len && d < D) {
            qv = Q[base + (long)s * D + d];
            kv = K[base + (long)s * D + d];
            vv = V[base + (long)s * D + d];
        }
        lds_q[s * DP + d] = qv;
        lds_k[s * DP + d] = kv;
        lds_vt[d * SP + s] = vv;
    }
    __syncthreads();

    // ---- scores: this wave's 16 rows x 64 cols, K-dim = D ---------------
    at_f32x4 sc[4];                            // [col_tile]
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) sc[ct] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < KP / 32; ++ks) {
        const int k0 = ks * 32 + fq * 8;
        const int arow = 16 * w + fr;
        at_bf16x8 a = *reinterpret_cast<const at_bf16x8*>(
            &lds_q[arow * DP + k0]);
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
            const int bcol = ct * 16 + fr;     // key index
            at_bf16x8 b = *reinterpret_cast<const at_bf16x8*>(
                &lds_k[bcol * DP + k0]);
            sc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, sc[ct], 0, 0, 0);
        }
    }

    // ---- row softmax on fragments --------------------------------------
    // lane holds, for each rr (4 rows), cols {16ct + fr}; a row's 64
    // entries live on the 16 lanes sharing fq. Mask cols >= s_len.
    float p[4][4];                              // [rr][ct]
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
        float m = -1e30f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
            float v = sc[ct][rr] * scale;
            if (ct * 16 + fr >= s_len) v = -1e30f;
            p[rr][ct] = v;
            m = fmaxf(m, v);
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
            m = fmaxf(m, __shfl_xor(m, off, 64));
        float sum = 0.0f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
            float e = __expf(p[rr][ct] - m);
            p[rr][ct] = e;
            sum += e;
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
            sum += __shfl_xor(sum, off, 64);
        const float inv = __builtin_amdgcn_rcpf(sum);   // 1-ulp; P is bf16 anyway
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
            const int row = 16 * w + fq * 4 + rr;
            const at_bf16 pv = (at_bf16)(p[rr][ct] * inv);
            lds_p[row * SP + ct * 16 + fr] = pv;
            if (P_out != nullptr)
                P_out[bh * (long)(ATT_S * ATT_S) + row * ATT_S
                      + ct * 16 + fr] = pv;
        }
    }
    __syncthreads();

    // ---- O = P @ V : rows [16w,16w+16) x D cols, K-dim = 64 -------------
    at_f32x4 acc[D / 16];
#pragma unroll
    for (int ct = 0; ct < D / 16; ++ct) acc[ct] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
        const int k0 = ks * 32 + fq * 8;
        const int arow = 16 * w + fr;
        at_bf16x8 a = *reinterpret_cast<const at_bf16x8*>(
            &lds_p[arow * SP + k0]);
#pragma unroll
        for (int ct = 0; ct < D / 16; ++ct) {
            const int bcol = ct * 16 + fr;     // output feature dim
            at_bf16x8 b = *reinterpret_cast<const at_bf16x8*>(
                &lds_vt[bcol * SP + k0]);
            acc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc[ct], 0, 0, 0);
        }
    }

#pragma unroll
    for (int ct = 0; ct < D / 16; ++ct) {
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
            const int row = 16 * w + fq * 4 + rr;
            const int col = ct * 16 + fr;
            if (row < s_len)
                O[base + (long)row * D + col] = (at_bf16)acc[ct][rr];
        }
    }
}


// ---------------------------------------------------------------------------
// Fused attention backward (training path): given the softmax matrix P
// saved by the forward, computes
//   dV = P^T dO      dP = dO V^T      dS = scale * P o (dP - rowsum(dPoP))
//   dQ = dS K        dK = dS^T Q
// All five GEMMs run on the same 16x16x32 bf16 MFMA fragments as the
// forward (C = A @ B^T with row-major A/B tiles); the softmax-jacobian
// rowsum is the same 16-lane shfl-xor reduction as the forward softmax.
// One block per (batch, head); transposed copies are staged in LDS.
// ---------------------------------------------------------------------------
template <int D>
__global__ void __launch_bounds__(256) attn_bwd_kernel(
    const at_bf16* __restrict__ Q,    // (BH, S, D)
    const at_bf16* __restrict__ K,
    const at_bf16* __restrict__ V,
    const at_bf16* __restrict__ P,    // (BH, 64, 64) from forward
    const at_bf16* __restrict__ dO,   // (BH, S, D)
    at_bf16* __restrict__ dQ,
    at_bf16* __restrict__ dK,
    at_bf16* __restrict__ dV,
    int s_len, float scale, long bh_count)
{
    constexpr int KP = (D < 32 ? 32 : D);
    constexpr int DP = KP + 8;
    constexpr int SP = ATT_S + 8;

    __shared__ at_bf16 lds_do[ATT_S * DP];     // dO rows
    __shared__ at_bf16 lds_dot[KP * SP];       // dO^T: [d][s]
    __shared__ at_bf16 lds_v[ATT_S * DP];      // V rows
    __shared__ at_bf16 lds_qt[KP * SP];        // Q^T: [d][s]
    __shared__ at_bf16 lds_kt[KP * SP];        // K^T: [d][s]
    __shared__ at_bf16 lds_p[ATT_S * SP];      // P rows
    __shared__ at_bf16 lds_pt[ATT_S * SP];     // P^T
    __shared__ at_bf16 lds_ds[ATT_S * SP];     // dS rows
    __shared__ at_bf16 lds_dst[ATT_S * SP];    // dS^T

    const long bh = blockIdx.x;
    if (bh >= bh_count) return;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;
    const int fr = lane & 15;
    const int fq = lane >> 4;

    const long base = bh * (long)s_len * D;
    for (int i = tid; i < ATT_S * KP; i += 256) {
        const int s = i / KP, d = i % KP;
        at_bf16 dov = (at_bf16)0.0f, vv = (at_bf16)0.0f;
        at_bf16 qv = (at_bf16)0.0f, kv = (at_bf16)0.0f;
        if (s < s_len && d < D) {
            dov = dO[base + (long)s * D + d];
            vv = V[base + (long)s * D + d];
            qv = Q[base + (long)s * D + d];
            kv = K[base + (long)s * D + d];
        }
        lds_do[s * DP + d] = dov;
        lds_dot[d * SP + s] = dov;
        lds_v[s * DP + d] = vv;
        lds_qt[d * SP + s] = qv;
        lds_kt[d * SP + s] = kv;
    }
    const long pbase = bh * (long)(ATT_S * ATT_S);
    for (int i = tid; i < ATT_S * ATT_S; i += 256) {
        const int r = i / ATT_S, c = i % ATT_S;
        const at_bf16 pv = P[pbase + i];
        lds_p[r * SP + c] = pv;
        lds_pt[c * SP + r] = pv;
    }
    __syncthreads();

    // ---- dP = dO @ V^T on this wave's 16 rows ---------------------------
    at_f32x4 dp[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) dp[ct] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < KP / 32; ++ks) {
        const int k0 = ks * 32 + fq * 8;
        const int arow = 16 * w + fr;
        at_bf16x8 a = *reinterpret_cast<const at_bf16x8*>(
            &lds_do[arow * DP + k0]);
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
            const int bcol = ct * 16 + fr;
            at_bf16x8 b = *reinterpret_cast<const at_bf16x8*>(
                &lds_v[bcol * DP + k0]);
            dp[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, dp[ct], 0, 0, 0);
        }
    }

    // ---- dS = scale * P o (dP - rowsum(dP o P)) -------------------------
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
        const int row = 16 * w + fq * 4 + rr;
        float pv[4], rs = 0.0f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
            pv[ct] = (float)lds_p[row * SP + ct * 16 + fr];
            rs += dp[ct][rr] * pv[ct];
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
            rs += __shfl_xor(rs, off, 64);
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
            const int col = ct * 16 + fr;
            const at_bf16 ds =
                (at_bf16)(scale * pv[ct] * (dp[ct][rr] - rs));
            lds_ds[row * SP + col] = ds;
            lds_dst[col * SP + row] = ds;
        }
    }
    __syncthreads();

    // ---- three output GEMMs (K-dim = 64 in two 32 chunks) ---------------
    at_f32x4 gq[D / 16], gk[D / 16], gv[D / 16];
#pragma unroll
    for (int ct = 0; ct < D / 16; ++ct) {
        gq[ct] = {0.f, 0.f, 0.f, 0.f};
        gk[ct] = {0.f, 0.f, 0.f, 0.f};
        gv[ct] = {0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
        const int k0 = ks * 32 + fq * 8;
        const int arow = 16 * w + fr;
        at_bf16x8 a_ds = *reinterpret_cast<const at_bf16x8*>(
            &lds_ds[arow * SP + k0]);
        at_bf16x8 a_dst = *reinterpret_cast<const at_bf16x8*>(
            &lds_dst[arow * SP + k0]);
        at_bf16x8 a_pt = *reinterpret_cast<const at_bf16x8*>(
            &lds_pt[arow * SP + k0]);
#pragma unroll
        for (int ct = 0; ct < D / 16; ++ct) {
            const int bcol = ct * 16 + fr;
            at_bf16x8 b_kt = *reinterpret_cast<const at_bf16x8*>(
                &lds_kt[bcol * SP + k0]);
            at_bf16x8 b_qt = *reinterpret_cast<const at_bf16x8*>(
                &lds_qt[bcol * SP + k0]);
            at_bf16x8 b_dot = *reinterpret_cast<const at_bf16x8*>(
                &lds_dot[bcol * SP + k0]);
            gq[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_ds, b_kt, gq[ct], 0, 0, 0);    // dQ = dS @ K
            gk[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_dst, b_qt, gk[ct], 0, 0, 0);   // dK = dS^T @ Q
            gv[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_pt, b_dot, gv[ct], 0, 0, 0);   // dV = P^T @ dO
        }
    }

#pragma unroll
    for (int ct = 0; ct < D / 16; ++ct) {
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
            const int row = 16 * w + fq * 4 + rr;
            const int col = ct * 16 + fr;
            if (row < s_len) {
                dQ[base + (long)row * D + col] = (at_bf16)gq[ct][rr];
                dK[base + (long)row * D + col] = (at_bf16)gk[ct][rr];
                dV[base + (long)row * D + col] = (at_bf16)gv[ct][rr];
            }
        }
    }
}

}  // namespace

extern "C" void launch_attn_fwd(const void* Q, const void* K, const void* V,
                                void* O, void* P_out, long bh_count,
                                int s_len, int d_head, float scale,
                                hipStream_t stream) {
    if (s_len > ATT_S)
        throw std::runtime_error("attn_fwd: s_len must be <= 64");
    dim3 grid((unsigned)bh_count), block(256);
#define ATT_CASE(DD)                                                         \
    case DD:                                                                 \
        hipLaunchKernelGGL(attn_fwd_kernel<DD>, grid, block, 0, stream,      \
                           (const at_bf16*)Q, (const at_bf16*)K,             \
                           (const at_bf16*)V, (at_bf16*)O,                   \
                           (at_bf16*)P_out, s_len, scale, bh_count);         \
        break;
    switch (d_head) {
        ATT_CASE(16)
        ATT_CASE(32)
        ATT_CASE(64)
        default:
            throw std::runtime_error("attn_fwd: d_head must be 16/32/64");
    }
#undef ATT_CASE
}

extern "C" void launch_attn_bwd(const void* Q, const void* K, const void* V,
                                const void* P, const void* dO, void* dQ,
                                void* dK, void* dV, long bh_count,
                                int s_len, int d_head, float scale,
                                hipStream_t stream) {
    if (s_len > ATT_S)
        throw std::runtime_error("attn_bwd: s_len must be <= 64");
    dim3 grid((unsigned)bh_count), block(256);
#define ATTB_CASE(DD)                                                        \
    case DD:                                                                 \
        hipLaunchKernelGGL(attn_bwd_kernel<DD>, grid, block, 0, stream,      \
                           (const at_bf16*)Q, (const at_bf16*)K,             \
                           (const at_bf16*)V, (const at_bf16*)P,             \
                           (const at_bf16*)dO, (at_bf16*)dQ, (at_bf16*)dK,   \
                           (at_bf16*)dV, s_len, scale, bh_count);            \
        break;
    switch (d_head) {
        ATTB_CASE(16)
        ATTB_CASE(32)
        ATTB_CASE(64)
        default:
            throw std::runtime_error("attn_bwd: d_head must be 16/32/64");
    }
#undef ATTB_CASE
}
