// Rolling technical-indicator kernels.
//
// Replaces the reference's per-update `ta`-library indicator pass
// (binance_ml_strategy.py:40-182, market_monitor_service.py:219-298) with a
// chunk-parallel scan: one lane owns a CHUNK of one symbol's candle stream
// and re-converges the EMA/Wilder recurrences over a WARM prefix before its
// output region (exponential forgetting: (1-2/(p+1))^512 underflows f32 for
// every p <= 64, so chunked output == sequential output). Rolling-window
// indicators (Bollinger, Stoch/Williams min-max, VWAP) use per-lane LDS
// rings and are exactly windowed.
//
// Output layout (nsym, T, NIND) f32:
//   0 ema12  1 ema26  2 macd  3 macd_signal  4 macd_hist  5 rsi14
//   6 bb_mid 7 bb_up  8 bb_lo 9 atr14  10 stoch_k  11 williams_r  12 vwap20
// CPU golden reference: ops/indicators.py.

#include "common.hpp"

#define IND_NIND 13
#define IND_CHUNK 2048
#define IND_WARM 512
#define IND_BLOCK 128
#define RING_W 20      // bollinger + vwap window
#define RING_S 14      // stoch/williams window
// per-lane ring floats: close 20 + high 14 + low 14 + pv 20 + vol 20 = 88
#define LANE_RING 89   // +1 pad -> odd stride, banks spread

namespace {

__global__ void __launch_bounds__(IND_BLOCK) indicators_kernel(
    const float* __restrict__ candles,   // (nsym, T, 4) [close,high,low,vol]
    float* __restrict__ out,             // (nsym, T, NIND)
    int nsym, int T, int nchunks)
{
    __shared__ float ring[IND_BLOCK * LANE_RING];
    const int lane_gid = blockIdx.x * blockDim.x + threadIdx.x;
    if (lane_gid >= nsym * nchunks) return;
    const int sym = lane_gid / nchunks;
    const int chunk = lane_gid % nchunks;
    const int t_out0 = chunk * IND_CHUNK;
    const int t_out1 = min(t_out0 + IND_CHUNK, T);
    const int t_start = max(0, t_out0 - IND_WARM);

    float* r_close = ring + threadIdx.x * LANE_RING;   // [20]
    float* r_high = r_close + RING_W;                  // [14]
    float* r_low = r_high + RING_S;                    // [14]
    float* r_pv = r_low + RING_S;                      // [20]
    float* r_vol = r_pv + RING_W;                      // [20]
    for (int i = 0; i < LANE_RING; ++i) r_close[i] = 0.0f;

    const float a12 = 2.0f / 13.0f, a26 = 2.0f / 27.0f, a9 = 2.0f / 10.0f;
    float ema12 = 0.f, ema26 = 0.f, sig = 0.f;
    float avg_gain = 0.f, avg_loss = 0.f, atr = 0.f;
    double bb_sum = 0.0, bb_sum2 = 0.0;   // f64: see engine_cpu.py rationale
    float pv_sum = 0.f, vol_sum = 0.f;
    float prev_close = 0.f;

    const float4* sc =
        reinterpret_cast<const float4*>(candles + (long)sym * T * 4);

    for (int t = t_start; t < t_out1; ++t) {
        const float4 c4 = sc[t];
        const float close = c4.x, high = c4.y, low = c4.z, vol = c4.w;
        const int steps = t - t_start;           // 0-based within this lane

        float change;
        if (steps == 0) {
            ema12 = close; ema26 = close; change = 0.0f;
            prev_close = close;
        } else {
            ema12 += a12 * (close - ema12);
            ema26 += a26 * (close - ema26);
            change = close - prev_close;
        }
        const float macd = ema12 - ema26;
        sig += a9 * (macd - sig);

        const float gain = fmaxf(change, 0.0f);
        const float lossv = fmaxf(-change, 0.0f);
        // reciprocal-multiply + v_rcp instead of IEEE div throughout this
        // kernel: the CPU reference divides, the GPU-vs-CPU test runs at
        // rtol 1e-3 (chunk-warmup reconvergence noise dominates 1-ulp rcp)
        avg_gain += (gain - avg_gain) * (1.0f / 14.0f);
        avg_loss += (lossv - avg_loss) * (1.0f / 14.0f);
        const float rsi = 100.0f * avg_gain *
            __builtin_amdgcn_rcpf(avg_gain + fmaxf(avg_loss, 1e-9f));

        const float tr = fmaxf(high - low,
                               fmaxf(fabsf(high - prev_close),
                                     fabsf(low - prev_close)));
        atr += (tr - atr) * (1.0f / 14.0f);
        prev_close = close;

        // Bollinger ring (window 20, f64 rolling sum/sumsq)
        {
            const int ri = steps % RING_W;
            const double old = (double)r_close[ri];
            const double c64 = (double)close;
            bb_sum += c64 - old;
            bb_sum2 += c64 * c64 - old * old;
            r_close[ri] = close;
        }
        // steps is the (wave-uniform) loop counter: the f64 divide only
        // executes during the first RING_W-1 iterations of each chunk
        double ibcnt = 1.0 / (double)RING_W;
        if (steps + 1 < RING_W) ibcnt = 1.0 / (steps + 1.0);
        const double mean64 = bb_sum * ibcnt;
        const double var64 = fmax(bb_sum2 * ibcnt - mean64 * mean64, 0.0);
        const float mean = (float)mean64;
        const float sd = __builtin_amdgcn_sqrtf((float)var64);

        // Stoch/Williams rings (window 14): scan for min/max
        {
            const int ri = steps % RING_S;
            r_high[ri] = high;
            r_low[ri] = low;
        }
        const int scnt = min(steps + 1, RING_S);
        float hmax = -1e30f, lmin = 1e30f;
        for (int i = 0; i < scnt; ++i) {
            hmax = fmaxf(hmax, r_high[i]);
            lmin = fminf(lmin, r_low[i]);
        }
        const float irng = __builtin_amdgcn_rcpf(fmaxf(hmax - lmin, 1e-9f));
        const float stoch_k = (close - lmin) * irng * 100.0f;
        const float williams = -100.0f * (hmax - close) * irng;

        // VWAP ring (window 20, typical price)
        {
            const int ri = steps % RING_W;
            const float tp = (high + low + close) * (1.0f / 3.0f);
            const float pv = tp * vol;
            pv_sum += pv - r_pv[ri];
            vol_sum += vol - r_vol[ri];
            r_pv[ri] = pv;
            r_vol[ri] = vol;
        }
        const float vwap = pv_sum * __builtin_amdgcn_rcpf(fmaxf(vol_sum, 1e-9f));

        if (t >= t_out0) {
            float* o = out + ((long)sym * T + t) * IND_NIND;
            o[0] = ema12; o[1] = ema26; o[2] = macd; o[3] = sig;
            o[4] = macd - sig; o[5] = rsi;
            o[6] = mean; o[7] = mean + 2.0f * sd; o[8] = mean - 2.0f * sd;
            o[9] = atr; o[10] = stoch_k; o[11] = williams; o[12] = vwap;
        }
    }
}

}  // namespace

extern "C" void launch_indicators(const float* candles, float* out, int nsym,
                                  int T, int nind, hipStream_t stream) {
    if (nind != IND_NIND)
        throw std::runtime_error("indicators: nind must be 13");
    int nchunks = (T + IND_CHUNK - 1) / IND_CHUNK;
    long lanes = (long)nsym * nchunks;
    dim3 grid((lanes + IND_BLOCK - 1) / IND_BLOCK);
    hipLaunchKernelGGL(indicators_kernel, grid, dim3(IND_BLOCK), 0, stream,
                       candles, out, nsym, T, nchunks);
}
