// GPU backtest engine — the headline kernel.
//
// Replaces the reference's per-candle Python loop
// (backtesting/strategy_tester.py:190-300, services/strategy_evaluation.py:777-878)
// with CDNA4 lanes marching candles:
//   - a block = 256 lanes; each lane simulates ILP param-sets of one
//     symbol. Measured: ILP=2 (two independent dependency chains/lane,
//     148 VGPR, 3 waves/SIMD) is 10% SLOWER than ILP=1 (96 VGPR, 5
//     waves/SIMD) — 323 vs 290 G candles/s at pop=1024 x 64 sym x 16
//     segments: wave-level latency hiding beats in-lane ILP here because
//     both chains stall on the same LDS broadcast + f64 sequence. ILP=1
//     is the default; BT_FORCE_ILP2=1 re-runs the experiment.
//   - candle tiles staged in LDS with a MAX_WIN-candle HALO: close history
//     is shared by every lane of the block, so the Bollinger "ring buffer"
//     is just a broadcast read of close[t-W] from the halo tile — no
//     per-lane LDS state at all (v1 used 32 KB/block of per-lane rings and
//     was LDS-occupancy-bound at 4 waves/SIMD)
//   - EMA/MACD/Wilder-RSI are O(1) register recurrences per candle
//   - the position state machine (SL/TP/trailing/vote exits — semantics of
//     trade_executor_service.py:55-399 + binance_ml_strategy.py:489-543)
//     matches backtesting/engine_cpu.py bit-for-bit (fp-contract off, same
//     operation order, f64 Bollinger sums both sides).
//
// Grid: nsym * ceil(P/(256*ILP)) blocks, XCD-affine map keeps the blocks
// of one symbol on one XCD so their shared candle stream stays in that
// XCD's L2.

#include "common.hpp"

#define BT_BLOCK 256
#define BT_TILE 256
#define BT_MAXWIN 32           // == strategy.py MAX_WIN
#define BT_HALO 64             // == strategy.py SHARED_HALO (covers sma50)
#define BT_SPAN (BT_TILE + BT_HALO)
#define BT_NPARAM 19
#define BT_NMETRIC 10
#define BT_WARMUP 128          // == strategy.py WARMUP
#define BT_RESNAP 16384        // == strategy.py RESNAP (BB sum resnap)
#define BT_FEE 0.001f
#define BT_EPS 1e-9f
// float32(sqrt(525600)) EXACTLY (0x44353ee6) — a shorter decimal literal
// rounds to the neighbor 0x44353ee5 and biases every sharpe by 1 ulp
#define BT_ANNUALIZE 0x1.6a7dccp+9f

namespace {

struct LaneParams {
    float inv_rsi_p, rsi_os, rsi_ob;
    float a_f, a_s, a_sig;
    int bb_w;
    float bb_k, bb_bth, bb_sth;
    int entry_v, exit_v;
    float size_pct, sl_pct, tp_pct, trail_pct, trail_act;
    float stoch_os, stoch_ob, will_os, will_ob;
};

// one strategy's full simulation state (indicators + position + metrics)
struct BtState {
    LaneParams q;
    float ema_f, ema_s, sig;
    float avg_gain, avg_loss;
    double bb_sum, bb_sum2, inv_w;    // f64 sums: f32 drifts + cancels
    float cash, units;
    bool in_pos;
    float entry_cost, entry_price;
    float stop, tp, peak;
    float equity, max_eq, max_dd;
    float n_trades, wins, gross_p, gross_l;
    float sum_ret, sum_ret2;

    __device__ void load(const float* __restrict__ pr, float initial_equity)
    {
        q.inv_rsi_p = 1.0f / fmaxf(floorf(pr[0]), 1.0f);
        q.rsi_os = pr[1]; q.rsi_ob = pr[2];
        q.a_f = 2.0f / (pr[3] + 1.0f);
        q.a_s = 2.0f / (pr[4] + 1.0f);
        q.a_sig = 2.0f / (pr[5] + 1.0f);
        q.bb_w = min(max((int)pr[6], 2), BT_MAXWIN);
        q.bb_k = pr[7]; q.bb_bth = pr[8]; q.bb_sth = pr[9];
        q.entry_v = (int)pr[10]; q.exit_v = (int)pr[11];
        q.size_pct = pr[12]; q.sl_pct = pr[13]; q.tp_pct = pr[14];
        q.trail_pct = pr[15]; q.trail_act = pr[16];
        q.stoch_os = pr[17]; q.stoch_ob = pr[18];
        q.will_os = q.stoch_os - 100.0f;
        q.will_ob = q.stoch_ob - 100.0f;
        ema_f = ema_s = sig = 0.f;
        avg_gain = avg_loss = 0.f;
        bb_sum = bb_sum2 = 0.0;
        inv_w = 1.0 / (double)q.bb_w;
        cash = initial_equity; units = 0.f;
        in_pos = false;
        entry_cost = entry_price = stop = tp = peak = 0.f;
        equity = initial_equity; max_eq = initial_equity; max_dd = 0.f;
        n_trades = wins = gross_p = gross_l = 0.f;
        sum_ret = sum_ret2 = 0.f;
    }

    // Drift-free Bollinger resnap (strategy.py RESNAP): recompute the
    // window sums directly, oldest->newest, from the close history.
    // `hist` points at close[t] (the resnap candle itself, j = 0 term).
    // (noinline was tried here: 86 VGPR/no spills but the call
    // inside the tile loop cost 25% — inlined + launch-bounds wins)
    __device__ void resnap(const float* __restrict__ hist)
    {
#pragma clang fp contract(off)
        double s = 0.0, s2 = 0.0;
        const int w = q.bb_w;
        for (int j = BT_MAXWIN - 1; j >= 0; --j) {
            if (j < w) {
                double c = (double)hist[-j];
                s += c;
                s2 += c * c;
            }
        }
        bb_sum = s;
        bb_sum2 = s2;
    }

    // SKIP_BB is a compile-time flag (a resnap() just replaced the
    // sums for this candle) so the steady-state loop carries no
    // per-candle branch — the resnap candle is peeled at the call site.
    template <bool SKIP_BB>
    __device__ void step(int t, float close, float high, float low,
                         float oldc, float change, float4 sv)
    {
#pragma clang fp contract(off)           // match the numpy f32 reference
        // --- 1. indicators -------------------------------------------
        if (t == 0) { ema_f = close; ema_s = close; }
        else {
            ema_f += q.a_f * (close - ema_f);
            ema_s += q.a_s * (close - ema_s);
        }
        float macd = ema_f - ema_s;
        sig += q.a_sig * (macd - sig);
        float macd_hist = macd - sig;

        float gain = fmaxf(change, 0.0f);
        float loss = fmaxf(-change, 0.0f);
        avg_gain += (gain - avg_gain) * q.inv_rsi_p;
        avg_loss += (loss - avg_loss) * q.inv_rsi_p;
        // division-free RSI votes (engine_cpu.py): rsi<thr <=>
        // 100*ag < thr*(ag+al')
        float rsi_num = 100.0f * avg_gain;
        float rsi_den = avg_gain + fmaxf(avg_loss, BT_EPS);

        // Bollinger: close[t - W] comes from the shared halo tile
        // (== the zero-initialized per-lane ring of engine_cpu.py)
        if (!SKIP_BB) {
            double old = (double)oldc;
            double c64 = (double)close;
            bb_sum += c64 - old;
            bb_sum2 += c64 * c64 - old * old;
        }
        double inv_cnt = inv_w;
        if (t < BT_MAXWIN && t + 1 < q.bb_w)   // uniformly skipped t>=32
            inv_cnt = 1.0 / (t + 1.0);
        double mean64 = bb_sum * inv_cnt;
        double var64 = fmax(bb_sum2 * inv_cnt - mean64 * mean64, 0.0);
        float mean = (float)mean64;
        float std_ = sqrtf((float)var64);
        float band = q.bb_k * std_;
        // division-free BB votes (engine_cpu.py): pos<thr <=> num<thr*den
        float bb_num = close - (mean - band);
        float bb_den = fmaxf(2.0f * band, BT_EPS);

        // --- 2. votes (6-indicator TradingSignal voting) -------------
        int net = 0;
        if (t >= BT_WARMUP) {
            int buy = (rsi_num < q.rsi_os * rsi_den) +
                      (macd_hist > 0.0f) +
                      (bb_num < q.bb_bth * bb_den) +
                      (sv.x < q.stoch_os * sv.z) +
                      (sv.y < q.will_os * sv.z) +
                      (sv.w > 0.0f);
            int sell = (rsi_num > q.rsi_ob * rsi_den) +
                       (macd_hist < 0.0f) +
                       (bb_num > q.bb_sth * bb_den) +
                       (sv.x > q.stoch_ob * sv.z) +
                       (sv.y > q.will_ob * sv.z) +
                       (sv.w < 0.0f);
            net = buy - sell;
        }

        // --- 3. position management ----------------------------------
        if (in_pos) {
            peak = fmaxf(peak, high);
            bool trail_on = (q.trail_pct > 0.0f) &&
                            (peak >= entry_price * (1.0f + q.trail_act));
            if (trail_on)
                stop = fmaxf(stop, peak * (1.0f - q.trail_pct));
            bool hit_sl = low <= stop;
            bool hit_tp = !hit_sl && high >= tp;
            bool hit_sig = !hit_sl && !hit_tp && net <= -q.exit_v;
            if (hit_sl || hit_tp || hit_sig) {
                float exit_price = hit_sl ? stop : (hit_tp ? tp : close);
                float proceeds = units * exit_price * (1.0f - BT_FEE);
                float pnl = proceeds - entry_cost;
                cash += proceeds;
                n_trades += 1.0f;
                wins += (pnl > 0.0f) ? 1.0f : 0.0f;
                gross_p += fmaxf(pnl, 0.0f);
                gross_l += fmaxf(-pnl, 0.0f);
                units = 0.0f;
                in_pos = false;
            }
        } else if (t >= BT_WARMUP && net >= q.entry_v) {
            float cost = fminf(q.size_pct * equity, cash);
            units = cost * (1.0f - BT_FEE) / close;
            cash -= cost;
            entry_cost = cost;
            entry_price = close;
            stop = close * (1.0f - q.sl_pct);
            tp = close * (1.0f + q.tp_pct);
            peak = close;
            in_pos = true;
        }

        // --- 4. mark to market ---------------------------------------
        // flat lanes: new_eq == cash == equity exactly, so r == 0 and
        // every accumulator is unchanged — skipping is bit-identical
        // to engine_cpu.py (which computes r = cash/cash - 1 = 0)
        if (units != 0.0f || cash != equity) {
            float new_eq = cash + units * close;
            float r = new_eq / equity - 1.0f;
            sum_ret += r;
            sum_ret2 += r * r;
            equity = new_eq;
            max_eq = fmaxf(max_eq, equity);
            max_dd = fmaxf(max_dd, (max_eq - equity) / max_eq);
        }
    }

    __device__ void finalize(float* __restrict__ out, int T) const
    {
#pragma clang fp contract(off)
        // (engine_cpu.finalize_metrics semantics)
        float n = (float)max(T, 1);
        float mean_r = sum_ret / n;
        float var_r = fmaxf(sum_ret2 / n - mean_r * mean_r, 0.0f);
        float sharpe = mean_r / fmaxf(sqrtf(var_r), BT_EPS) * BT_ANNUALIZE;
        if (!(n_trades > 0.0f)) sharpe = 0.0f;
        float win_rate = wins / fmaxf(n_trades, 1.0f);
        float fitness = (n_trades > 0.0f)
                            ? sharpe + win_rate - 2.0f * max_dd
                            : -1.0f;
        out[0] = equity; out[1] = n_trades; out[2] = wins;
        out[3] = gross_p; out[4] = gross_l; out[5] = max_dd;
        out[6] = sum_ret; out[7] = sum_ret2;
        out[8] = sharpe; out[9] = fitness;
    }
};

template <int ILP, bool HAS_RESNAP>
// second arg: min resident blocks/CU — pins the allocation at <=102
// VGPR so 5 waves/SIMD stay resident (the resnap addition had silently
// inflated the allocation to 160 VGPR = 3 waves/SIMD, a 20% headline
// regression; measured: occupancy beats spill-free codegen here)
__global__ void __launch_bounds__(BT_BLOCK, ILP == 1 ? 5 : 3) backtest_kernel(
    const float* __restrict__ candles,   // (nsym, T, 4)
    const float* __restrict__ pop,       // (P, NPARAM)
    float* __restrict__ metrics,         // (P, nsym, NMETRIC)
    int nsym, int T, int P, int chunks_per_sym, float initial_equity)
{
#pragma clang fp contract(off)           // match the numpy f32 reference
    __shared__ float chist[BT_SPAN];     // close history incl. halo
    __shared__ float hl[BT_SPAN][2];     // high, low
    // param-independent shared series, computed cooperatively per tile
    // (strategy.py step-1 spec: every lane of a symbol shares them)
    // packed per-candle shared vote inputs: {st_num, wl_num, srange,
    // trend} — st_num = 100*(close-lmin14), wl_num = -100*(hmax14-close),
    // srange = max(hmax-lmin, eps), trend = +-1/0 from close vs sma20/50.
    // All lane-independent, so one b128 broadcast read per candle.
    __shared__ float4 sh_vote[BT_TILE];

    // block -> (symbol, param chunk); same-symbol blocks share an XCD when
    // the shape allows (dispatcher places block b on XCD b%8).
    int bid = blockIdx.x;
    int sym, chunk;
    int nblocks = nsym * chunks_per_sym;
    if ((nsym & 7) == 0 && (nblocks & 7) == 0) {
        int xcd = bid & 7, j = bid >> 3;
        sym = xcd + 8 * (j / chunks_per_sym);
        chunk = j % chunks_per_sym;
    } else {
        sym = bid / chunks_per_sym;
        chunk = bid % chunks_per_sym;
    }

    const int tid = threadIdx.x;
    // lane tid simulates params {base + tid, base + tid + 256, ...}
    const int pbase = chunk * BT_BLOCK * ILP + tid;

    BtState st[ILP];
    bool act[ILP];
#pragma unroll
    for (int i = 0; i < ILP; ++i) {
        const int p = pbase + i * BT_BLOCK;
        act[i] = p < P;
        st[i].load(pop + (long)(act[i] ? p : 0) * BT_NPARAM,
                   initial_equity);
    }

    float prev_close = 0.f;
    const float4* sym_candles =
        reinterpret_cast<const float4*>(candles + (long)sym * T * 4);

    for (int t0 = 0; t0 < T; t0 += BT_TILE) {
        __syncthreads();
        // stage [t0 - HALO, t0 + TILE) with zero-fill left of t=0
        for (int i = tid; i < BT_SPAN; i += BT_BLOCK) {
            const int t = t0 - BT_HALO + i;
            if (t >= 0 && t < T) {
                float4 c = sym_candles[t];
                chist[i] = c.x;
                hl[i][0] = c.y;
                hl[i][1] = c.z;
            } else {
                chist[i] = 0.0f;
                hl[i][0] = 0.0f;
                hl[i][1] = 0.0f;
            }
        }
        __syncthreads();
        const int tend = min(BT_TILE, T - t0);
        // BB resnap at RESNAP-aligned tiles (engine_cpu.py lockstep):
        // window closes for candle t0 are chist[BT_HALO - j], j < bb_w.
        const bool resnap_tile = HAS_RESNAP && (t0 > 0) &&
                                 ((t0 & (BT_RESNAP - 1)) == 0);
        if (resnap_tile) {
#pragma unroll
            for (int i = 0; i < ILP; ++i)
                st[i].resnap(&chist[BT_HALO]);
        }
        // cooperative shared-series precompute: one thread per tile candle
        if (tid < tend) {
            const int t = t0 + tid;
            const int base = tid + BT_HALO;
            const float cl = chist[base];
            const int L14 = min(t + 1, 14);
            float hmax = -1e30f, lmin = 1e30f;
            for (int j = 0; j < L14; ++j) {
                hmax = fmaxf(hmax, hl[base - j][0]);
                lmin = fminf(lmin, hl[base - j][1]);
            }
            const int L20 = min(t + 1, 20);
            double s20 = 0.0;
            for (int j = 0; j < L20; ++j) s20 += (double)chist[base - j];
            const float sma20 = (float)(s20 / (double)L20);
            const int L50 = min(t + 1, 50);
            double s50 = 0.0;
            for (int j = 0; j < L50; ++j) s50 += (double)chist[base - j];
            const float sma50 = (float)(s50 / (double)L50);
            const float trend = (cl > sma20 && sma20 > sma50) ? 1.0f
                                : ((cl < sma20 && sma20 < sma50) ? -1.0f
                                                                 : 0.0f);
            sh_vote[tid] = make_float4(
                100.0f * (cl - lmin), -100.0f * (hmax - cl),
                fmaxf(hmax - lmin, BT_EPS), trend);
        }
        __syncthreads();
        int tt0 = 0;
        if (resnap_tile) {
            // peeled resnap candle: BB increment skipped (template
            // flag -> zero cost in the steady-state loop below)
            const float close = chist[BT_HALO];
            const float change = close - prev_close;
            const float4 sv = sh_vote[0];
#pragma unroll
            for (int i = 0; i < ILP; ++i)
                st[i].template step<true>(t0, close, hl[BT_HALO][0],
                                 hl[BT_HALO][1],
                                 chist[BT_HALO - st[i].q.bb_w],
                                 change, sv);
            prev_close = close;
            tt0 = 1;
        }
        for (int tt = tt0; tt < tend; ++tt) {
            const int t = t0 + tt;
            const float close = chist[tt + BT_HALO];
            const float high = hl[tt + BT_HALO][0];
            const float low = hl[tt + BT_HALO][1];
            const float change = (t == 0) ? 0.0f : close - prev_close;
            const float4 sv = sh_vote[tt];   // b128 broadcast
#pragma unroll
            for (int i = 0; i < ILP; ++i)
                st[i].template step<false>(t, close, high, low,
                                  chist[tt + BT_HALO - st[i].q.bb_w],
                                  change, sv);
            prev_close = close;
        }
    }

#pragma unroll
    for (int i = 0; i < ILP; ++i) {
        const int p = pbase + i * BT_BLOCK;
        if (act[i])
            st[i].finalize(metrics + ((long)p * nsym + sym) * BT_NMETRIC,
                           T);
    }
}

}  // namespace

extern "C" void launch_backtest(const float* candles, const float* pop,
                                float* metrics, int nsym, int T, int P,
                                float initial_equity, hipStream_t stream) {
    // ILP=1 is the measured winner (see header); BT_FORCE_ILP2=1 opts
    // into the 2-chain variant for perf A/Bs on future hardware.
    static const bool force2 = [] {
        const char* e = getenv("BT_FORCE_ILP2");
        return e && e[0] == '1';
    }();
    // histories short enough never to cross a RESNAP boundary run the
    // resnap-free instantiation: identical codegen (96 VGPR, 5 waves/
    // SIMD) to a kernel without the feature — the seg-64 headline path
    const bool rs = T > BT_RESNAP;
    if (force2 && P % (BT_BLOCK * 2) == 0) {
        int chunks = P / (BT_BLOCK * 2);
        auto k = rs ? backtest_kernel<2, true> : backtest_kernel<2, false>;
        hipLaunchKernelGGL(k, dim3(nsym * chunks),
                           dim3(BT_BLOCK), 0, stream,
                           candles, pop, metrics, nsym, T, P, chunks,
                           initial_equity);
    } else {
        int chunks = (P + BT_BLOCK - 1) / BT_BLOCK;
        auto k = rs ? backtest_kernel<1, true> : backtest_kernel<1, false>;
        hipLaunchKernelGGL(k, dim3(nsym * chunks),
                           dim3(BT_BLOCK), 0, stream,
                           candles, pop, metrics, nsym, T, P, chunks,
                           initial_equity);
    }
}
