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
// Backward: models/attention.py recomputes through torch autograd (the
// fused forward is the inference/serving path and the training forward;
// grads come from a differentiable recompute).
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
            lds_p[row * SP + ct * 16 + fr] = (at_bf16)(p[rr][ct] * inv);
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

}  // namespace

extern "C" void launch_attn_fwd(const void* Q, const void* K, const void* V,
                                void* O, long bh_count, int s_len,
                                int d_head, float scale,
                                hipStream_t stream) {
    if (s_len > ATT_S)
        throw std::runtime_error("attn_fwd: s_len must be <= 64");
    dim3 grid((unsigned)bh_count), block(256);
#define ATT_CASE(DD)                                                         \
    case DD:                                                                 \
        hipLaunchKernelGGL(attn_fwd_kernel<DD>, grid, block, 0, stream,      \
                           (const at_bf16*)Q, (const at_bf16*)K,             \
                           (const at_bf16*)V, (at_bf16*)O, s_len, scale,     \
                           bh_count);                                        \
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
