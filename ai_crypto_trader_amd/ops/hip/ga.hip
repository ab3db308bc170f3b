// GPU-resident genetic-algorithm evolution ops.
//
// Replaces services/genetic_algorithm.py's host-side tournament selection +
// elitism (:135-161), uniform crossover (:163-189) and int/float mutation
// (:191-223) with one elementwise kernel over the device-resident
// population: thread = one child individual. Fitness evaluation itself is
// the backtest kernel; host only argsorts the (tiny) fitness vector for the
// elite indices. Population stays in HBM across generations; with DP
// sharding each rank evolves its own shard after an RCCL all-gather of
// fitness (parallel/dist.py).

#include "common.hpp"

#define GA_NPARAM 19

namespace {

__global__ void ga_evolve_kernel(
    const float* __restrict__ pop,      // (P, NPARAM) current generation
    const float* __restrict__ fitness,  // (P,)
    const int* __restrict__ order,      // (P,) fitness argsort, best first
    const float* __restrict__ bounds,   // (NPARAM, 3) lo, hi, is_int
    float* __restrict__ out,            // (P, NPARAM) next generation
    int P, int elite_k, int tournament, float cx_rate, float mut_rate,
    float mut_scale, uint64_t seed, uint64_t gen)
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= P) return;

    if (i < elite_k) {                       // elitism: copy the best as-is
        int src = order[i];
        for (int j = 0; j < GA_NPARAM; ++j)
            out[i * GA_NPARAM + j] = pop[src * GA_NPARAM + j];
        return;
    }

    // Two tournaments of `tournament` contestants each.
    uint64_t ctr = ((uint64_t)i << 20) | (gen & 0xFFFFF);
    Philox4 r0 = philox4x32(seed, ctr, 0);
    Philox4 r1 = philox4x32(seed, ctr, 1);
    uint32_t rnd[8] = {r0.x, r0.y, r0.z, r0.w, r1.x, r1.y, r1.z, r1.w};
    int pa = -1, pb = -1;
    float fa = -1e30f, fb = -1e30f;
    for (int k = 0; k < tournament; ++k) {
        int c = (int)(rnd[k & 7] % (uint32_t)P);
        if (fitness[c] > fa) { fa = fitness[c]; pa = c; }
        int c2 = (int)(rnd[(k + 4) & 7] % (uint32_t)P);
        if (fitness[c2] > fb) { fb = fitness[c2]; pb = c2; }
    }

    // Uniform crossover + gaussian mutation, clipped to bounds.
    for (int j = 0; j < GA_NPARAM; ++j) {
        Philox4 rj = philox4x32(seed, ctr, 2 + (uint64_t)j);
        float u_cx = u32_to_unit(rj.x);
        float u_mut = u32_to_unit(rj.y);
        float v = (u_cx < cx_rate) ? pop[pa * GA_NPARAM + j]
                                   : pop[pb * GA_NPARAM + j];
        float lo = bounds[j * 3 + 0];
        float hi = bounds[j * 3 + 1];
        bool is_int = bounds[j * 3 + 2] > 0.0f;
        if (u_mut < mut_rate) {
            float2 g = box_muller(rj.z, rj.w);
            v += g.x * mut_scale * (hi - lo);
        }
        v = fminf(fmaxf(v, lo), hi);
        if (is_int) v = rintf(v);
        out[i * GA_NPARAM + j] = v;
    }
    // keep ema_slow > ema_fast (strategy.py clip_params invariant)
    if (out[i * GA_NPARAM + 4] <= out[i * GA_NPARAM + 3])
        out[i * GA_NPARAM + 4] = out[i * GA_NPARAM + 3] + 1.0f;
}

}  // namespace

extern "C" void launch_ga_evolve(const float* pop, const float* fitness,
                                 const int* order, const float* bounds,
                                 float* out, int P, int elite_k,
                                 int tournament, float cx_rate, float mut_rate,
                                 float mut_scale, uint64_t seed, uint64_t gen,
                                 hipStream_t stream) {
    dim3 grid((P + 255) / 256);
    hipLaunchKernelGGL(ga_evolve_kernel, grid, dim3(256), 0, stream, pop,
                       fitness, order, bounds, out, P, elite_k, tournament,
                       cx_rate, mut_rate, mut_scale, seed, gen);
}
