// Common device utilities for the MI355X (gfx950) kernels.
// CDNA4-only: wave64, no CUDA-compat paths.
#pragma once

#include <hip/hip_runtime.h>

#define DEV_INLINE __device__ __forceinline__

constexpr int WAVE = 64;   // CDNA wavefront width (gfx950)

// ---------------------------------------------------------------------------
// Philox4x32-7 counter-based RNG (Salmon et al. 2011). Stateless: the
// (seed, counter) pair fully determines the output, so every lane of every
// kernel draws independent, reproducible streams — used by the Monte-Carlo
// path generator (replacing monte_carlo_service.py:264-273's
// np.random.standard_normal) and the GA mutation ops
// (genetic_algorithm.py:191-223 semantics).
// 7 rounds: the Random123 paper's BigCrush-passing round count (10 is the
// library default safety margin). The MC pathgen is RNG-issue-bound on
// MI355X (PMC: 75 VALU instrs per MFMA instr), so the 3 saved rounds are
// ~10% of the whole kernel. ops/montecarlo.py's numpy twin uses the same
// count — change them together.
// ---------------------------------------------------------------------------
constexpr int PHILOX_ROUNDS = 7;
struct Philox4 {
    uint32_t x, y, z, w;
};

DEV_INLINE uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hi) {
    uint64_t p = (uint64_t)a * (uint64_t)b;
    *hi = (uint32_t)(p >> 32);
    return (uint32_t)p;
}

DEV_INLINE Philox4 philox4x32(uint64_t seed, uint64_t ctr_lo, uint64_t ctr_hi) {
    constexpr uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
    constexpr uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
    uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
    uint32_t c0 = (uint32_t)ctr_lo, c1 = (uint32_t)(ctr_lo >> 32);
    uint32_t c2 = (uint32_t)ctr_hi, c3 = (uint32_t)(ctr_hi >> 32);
#pragma unroll
    for (int r = 0; r < PHILOX_ROUNDS; ++r) {
        uint32_t hi0, hi1;
        uint32_t lo0 = mulhilo(M0, c0, &hi0);
        uint32_t lo1 = mulhilo(M1, c2, &hi1);
        uint32_t n0 = hi1 ^ c1 ^ k0;
        uint32_t n1 = lo1;
        uint32_t n2 = hi0 ^ c3 ^ k1;
        uint32_t n3 = lo0;
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
        k0 += W0; k1 += W1;
    }
    return {c0, c1, c2, c3};
}

// Uniform (0, 1]: never returns 0 so log() in Box-Muller is safe.
DEV_INLINE float u32_to_unit(uint32_t v) {
    return ((float)v + 1.0f) * 2.3283064e-10f;   // (v+1) * 2^-32
}

// Box-Muller: two N(0,1) from two U(0,1].
// Instruction-minimal for the RNG-issue-bound MC pathgen:
//  - V_SIN/V_COS_F32 natively compute sin/cos(2*pi*x) ("turns" input), so
//    the uniform u2 in (0,1] is EXACTLY the hardware domain: no 2*pi
//    multiply, no range reduction (__sincosf emits a 1/2pi mul + guards).
//  - -2*ln(u) = -2ln2 * log2(u): one fused constant on the raw v_log_f32.
//  - raw v_sqrt_f32 (1 ulp): libm sqrtf emits a ~10-instr scale/Newton
//    adjust sequence per call for correct rounding we don't need here.
// The numpy twin (ops/montecarlo.py) keeps sin(2*pi*u)/cos/log — identical
// math, transcendental rounding differs by ulps (tests use tolerances).
DEV_INLINE float2 box_muller(uint32_t a, uint32_t b) {
    float u1 = u32_to_unit(a);
    float u2 = u32_to_unit(b);
    float r = __builtin_amdgcn_sqrtf(
        -1.3862943611f * __builtin_amdgcn_logf(u1));
    float s = __builtin_amdgcn_sinf(u2);
    float c = __builtin_amdgcn_cosf(u2);
    return make_float2(r * c, r * s);
}

// Four N(0,1) draws from one Philox call.
DEV_INLINE float4 philox_normal4(uint64_t seed, uint64_t ctr_lo,
                                 uint64_t ctr_hi) {
    Philox4 p = philox4x32(seed, ctr_lo, ctr_hi);
    float2 ab = box_muller(p.x, p.y);
    float2 cd = box_muller(p.z, p.w);
    return make_float4(ab.x, ab.y, cd.x, cd.y);
}

// ---------------------------------------------------------------------------
// Wave / block reductions (64-lane wavefront).
// ---------------------------------------------------------------------------
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE);
    return v;   // valid in lane 0 of the wave
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_down(v, off, WAVE));
    return v;
}

DEV_INLINE float wave_reduce_min(float v) {
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v = fminf(v, __shfl_down(v, off, WAVE));
    return v;
}

// Block reduction through LDS; `scratch` needs blockDim.x/WAVE floats.
template <typename Op>
DEV_INLINE float block_reduce(float v, float* scratch, Op op, float ident) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v = op(v, __shfl_down(v, off, WAVE));
    if (lane == 0) scratch[wid] = v;
    __syncthreads();
    v = (threadIdx.x < nw) ? scratch[threadIdx.x] : ident;
    if (wid == 0) {
#pragma unroll
        for (int off = WAVE / 2; off > 0; off >>= 1)
            v = op(v, __shfl_down(v, off, WAVE));
    }
    return v;   // valid in thread 0
}

// ---------------------------------------------------------------------------
// XCD-aware block remap (8 XCDs, each with a private L2): keep blocks that
// share input panels on one XCD. Bijective for any nwg
// (cdna_hip_programming.md §5 "XCD swizzle must be bijective").
// ---------------------------------------------------------------------------
DEV_INLINE int xcd_swizzle(int bid, int nwg) {
    constexpr int NXCD = 8;
    if (nwg < NXCD) return bid;
    int q = nwg / NXCD, r = nwg % NXCD;
    int xcd = bid % NXCD, idx = bid / NXCD;
    return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

#define HIP_CHECK_LAST()                                                      \
    do {                                                                      \
        hipError_t e_ = hipGetLastError();                                    \
        if (e_ != hipSuccess)                                                 \
            throw std::runtime_error(std::string("HIP error: ") +             \
                                     hipGetErrorString(e_));                  \
    } while (0)
