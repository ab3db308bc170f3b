// Time-parallel continuous backtest — the seg=1 fast path.
//
// The classic kernel (backtest.hip) marches one lane per (param x symbol)
// through all T candles. At the flagship shape (pop=1024 x 64 symbols)
// that is 65,536 lanes = 1,024 waves = exactly 1 wave/SIMD on MI355X:
// the per-candle dependency chain (EMA/RSI/f64-Bollinger -> votes ->
// position machine) is fully latency-exposed and the launch runs ~2.7x
// below the chip's issue wall (round-1 PMC analysis,
// profiles/backtest_seg64_pmc.json).
//
// This file splits the problem into two kernels so the expensive part
// fills the chip:
//
//   1. bt_flags_kernel — indicators + votes only, time-split into S
//      shards per lane (S x more waves -> 4-8 waves/SIMD). Exactness:
//        - EMA/MACD-signal/Wilder-RSI are contractive recurrences; each
//          shard warm-starts them TAIL (default 2048) candles early,
//          which drives the difference from the continuous trajectory
//          below half an f32 ulp (worst contraction 1-1/64 per candle:
//          0.984^2048 ~ 1e-14) — bitwise reconvergence in practice,
//          asserted by tests/test_gpu_kernels GPU parity tests.
//        - Bollinger f64 rolling sums are NOT contractive; instead both
//          sequential engines resnap them from the raw window every
//          RESNAP=16384 candles (strategy.py), and shard bodies start on
//          RESNAP-aligned boundaries, so the shard's BB state is
//          *bit-identical* to the continuous run by construction.
//        - stoch/Williams/trend vote inputs are finite-window shared
//          series recomputed from the tile halo — exact by construction.
//      Output: per-lane entry/exit flag bitmaps (net >= entry_votes,
//      net <= -exit_votes), 1 bit per candle packed into uint64 words.
//
//   2. bt_trades_kernel — the exact position state machine + equity
//      accounting (identical op order to engine_cpu.py), reading flags
//      instead of recomputing votes. Still 1 wave/SIMD, but the
//      per-candle chain is ~5x shorter (no indicators, no f64, no
//      sqrt), flat candles are provably no-ops (equity==cash -> r==0,
//      engine_cpu semantics), and whole 64-candle words are skipped
//      when no lane of the wave is in a position and no entry flag is
//      set (__ballot).
//
// Flag layout: [sym][word][param] (uint64), word = t/64 — phase 1
// writes and phase 2 reads are both coalesced across the param axis.
//
// Reference semantics: backtesting/strategy.py (shared CPU/GPU
// contract); replaces the per-candle loop of the reference's
// backtesting/strategy_tester.py:190-300 at continuous (multi-year
// single-history) shape.

#include "common.hpp"

#define BT_BLOCK 256
#define BT_TILE 256
#define BT_MAXWIN 32
#define BT_HALO 64
#define BT_SPAN (BT_TILE + BT_HALO)
#define BT_NPARAM 19
#define BT_NMETRIC 10
#define BT_WARMUP 128
#define BT_RESNAP 16384
#define BT_FEE 0.001f
#define BT_EPS 1e-9f
// float32(sqrt(525600)) EXACTLY (0x44353ee6) — see backtest.hip
#define BT_ANNUALIZE 0x1.6a7dccp+9f

namespace {

// ---------------------------------------------------------------------
// Phase 1: indicator/vote state (no position, no metrics)
// ---------------------------------------------------------------------
struct FlagState {
    float inv_rsi_p, rsi_os, rsi_ob;
    float a_f, a_s, a_sig;
    int bb_w;
    float bb_k, bb_bth, bb_sth;
    int entry_v, exit_v;
    float stoch_os, stoch_ob, will_os, will_ob;

    float ema_f, ema_s, sig;
    float avg_gain, avg_loss;
    double bb_sum, bb_sum2, inv_w;

    __device__ void load(const float* __restrict__ pr)
    {
        inv_rsi_p = 1.0f / fmaxf(floorf(pr[0]), 1.0f);
        rsi_os = pr[1]; rsi_ob = pr[2];
        a_f = 2.0f / (pr[3] + 1.0f);
        a_s = 2.0f / (pr[4] + 1.0f);
        a_sig = 2.0f / (pr[5] + 1.0f);
        bb_w = min(max((int)pr[6], 2), BT_MAXWIN);
        bb_k = pr[7]; bb_bth = pr[8]; bb_sth = pr[9];
        entry_v = (int)pr[10]; exit_v = (int)pr[11];
        stoch_os = pr[17]; stoch_ob = pr[18];
        will_os = stoch_os - 100.0f;
        will_ob = stoch_ob - 100.0f;
        ema_f = ema_s = sig = 0.f;
        avg_gain = avg_loss = 0.f;
        bb_sum = bb_sum2 = 0.0;
        inv_w = 1.0 / (double)bb_w;
    }

    // warm-tail march: the same f32 recurrence ops as step(), minus
    // Bollinger (resnapped at body start) and votes
    __device__ void warm_step(float close, float change)
    {
#pragma clang fp contract(off)
        ema_f += a_f * (close - ema_f);
        ema_s += a_s * (close - ema_s);
        float macd = ema_f - ema_s;
        sig += a_sig * (macd - sig);
        float gain = fmaxf(change, 0.0f);
        float loss = fmaxf(-change, 0.0f);
        avg_gain += (gain - avg_gain) * inv_rsi_p;
        avg_loss += (loss - avg_loss) * inv_rsi_p;
    }

    __device__ void resnap(const float* __restrict__ hist)
    {
#pragma clang fp contract(off)
        double s = 0.0, s2 = 0.0;
        const int w = bb_w;
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

    // Full body candle: indicators + votes -> (entry, exit) bits.
    // SKIP_BB is compile-time (resnap peeled at the call site) so the
    // steady-state loop carries no per-candle branch.
    template <bool SKIP_BB>
    __device__ uint2 candle(int t, float close, float change,
                            float oldc, double csq_new, double csq_old,
                            float4 sv)
    {
#pragma clang fp contract(off)
        if (t == 0) { ema_f = close; ema_s = close; }
        else {
            ema_f += a_f * (close - ema_f);
            ema_s += a_s * (close - ema_s);
        }
        float macd = ema_f - ema_s;
        sig += a_sig * (macd - sig);
        float macd_hist = macd - sig;

        float gain = fmaxf(change, 0.0f);
        float loss = fmaxf(-change, 0.0f);
        avg_gain += (gain - avg_gain) * inv_rsi_p;
        avg_loss += (loss - avg_loss) * inv_rsi_p;
        float rsi_num = 100.0f * avg_gain;
        float rsi_den = avg_gain + fmaxf(avg_loss, BT_EPS);

        if (!SKIP_BB) {
            double old = (double)oldc;
            double c64 = (double)close;
            bb_sum += c64 - old;
            bb_sum2 += csq_new - csq_old;
        }
        double inv_cnt = inv_w;
        // the early-window divisor only exists in the very first tile
        // (t < MAX_WIN <= 32 < BT_TILE)
        if (t < BT_MAXWIN && t + 1 < bb_w)
            inv_cnt = 1.0 / (t + 1.0);
        double mean64 = bb_sum * inv_cnt;
        double var64 = fmax(bb_sum2 * inv_cnt - mean64 * mean64, 0.0);
        float mean = (float)mean64;
        float std_ = sqrtf((float)var64);
        float band = bb_k * std_;
        float bb_num = close - (mean - band);
        float bb_den = fmaxf(2.0f * band, BT_EPS);

        int net = 0;
        if (t >= BT_WARMUP) {
            int buy = (rsi_num < rsi_os * rsi_den) +
                      (macd_hist > 0.0f) +
                      (bb_num < bb_bth * bb_den) +
                      (sv.x < stoch_os * sv.z) +
                      (sv.y < will_os * sv.z) +
                      (sv.w > 0.0f);
            int sell = (rsi_num > rsi_ob * rsi_den) +
                       (macd_hist < 0.0f) +
                       (bb_num > bb_sth * bb_den) +
                       (sv.x > stoch_ob * sv.z) +
                       (sv.y > will_ob * sv.z) +
                       (sv.w < 0.0f);
            net = buy - sell;
        }
        return make_uint2(net >= entry_v, net <= -exit_v);
    }
};

// ---------------------------------------------------------------------
// Phase 2: position machine + metrics (exact engine_cpu op order).
//
// Runs at 1 wave/SIMD (65,536 lanes at the flagship shape), so the
// per-candle DEPENDENCY CHAIN is everything: the body is branchless
// (selects, like the numpy engine itself — which computes every
// quantity for every lane and np.where-selects) except the rare entry
// block, whose division would otherwise be a third serial IEEE-divide
// per candle. The two mark divisions (r and drawdown) only feed
// accumulators, never the trading state, so with a branchless body the
// scheduler pipelines them across unrolled iterations.
// ---------------------------------------------------------------------
struct TradeState {
    float size_pct;
    float fee_m;       // 1 - FEE
    float sl_m;        // 1 - stop_loss_pct
    float tp_m;        // 1 + take_profit_pct
    float trail_m;     // 1 - trailing_stop_pct
    float act_m;       // 1 + trailing_act_pct
    bool trail_en;     // trailing_stop_pct > 0
    float cash, units;
    bool in_pos;
    float entry_cost;
    float trail_arm;   // entry_price * act_m (armed threshold)
    float stop, tp, peak;
    float equity;
    float n_trades, wins, gross_p, gross_l;

    __device__ void load(const float* __restrict__ pr, float initial_equity)
    {
#pragma clang fp contract(off)
        size_pct = pr[12];
        fee_m = 1.0f - BT_FEE;
        sl_m = 1.0f - pr[13];
        tp_m = 1.0f + pr[14];
        trail_m = 1.0f - pr[15];
        act_m = 1.0f + pr[16];
        trail_en = pr[15] > 0.0f;
        cash = initial_equity; units = 0.f;
        in_pos = false;
        entry_cost = trail_arm = 0.f;
        stop = tp = peak = 0.f;
        equity = initial_equity;
        n_trades = wins = gross_p = gross_l = 0.f;
    }

    // bitwise state carry between chunked launches (f32 registers
    // stored/reloaded verbatim; in_pos as 0/1). Producer owns slots
    // [0..12]; the consumer wave owns [13..16] (max_eq, max_dd,
    // sum_ret, sum_ret2) — disjoint, saved by different threads.
    __device__ void save(float* __restrict__ s) const
    {
        s[0] = cash; s[1] = units; s[2] = in_pos ? 1.0f : 0.0f;
        s[3] = entry_cost; s[4] = trail_arm;
        s[5] = stop; s[6] = tp; s[7] = peak;
        s[8] = equity;
        s[9] = n_trades; s[10] = wins;
        s[11] = gross_p; s[12] = gross_l;
    }

    __device__ void restore(const float* __restrict__ s)
    {
        cash = s[0]; units = s[1]; in_pos = s[2] != 0.0f;
        entry_cost = s[3]; trail_arm = s[4];
        stop = s[5]; tp = s[6]; peak = s[7];
        equity = s[8];
        n_trades = s[9]; wins = s[10];
        gross_p = s[11]; gross_l = s[12];
    }

    // producer half of the per-candle step: position management + the
    // equity mark (engine_cpu sections 3 + the new_eq assignment of 4).
    // Returns new_eq for the consumer wave, which owns the r/drawdown
    // divisions and the return accumulators.
    __device__ float pstep_pos(float close, float high, float low,
                               bool ebit, bool xbit)
    {
#pragma clang fp contract(off)
        const bool pos0 = in_pos;
        float peak2 = fmaxf(peak, high);
        peak = pos0 ? peak2 : peak;
        bool trail_on = pos0 && trail_en && (peak >= trail_arm);
        float stop2 = fmaxf(stop, peak * trail_m);
        stop = trail_on ? stop2 : stop;

        bool hit_sl = pos0 && (low <= stop);
        bool hit_tp = pos0 && !hit_sl && (high >= tp);
        bool hit_sig = pos0 && !hit_sl && !hit_tp && xbit;
        bool exiting = hit_sl || hit_tp || hit_sig;
        float exit_price = hit_sl ? stop : (hit_tp ? tp : close);
        float proceeds = units * exit_price * fee_m;
        float pnl = proceeds - entry_cost;
        cash = exiting ? cash + proceeds : cash;
        n_trades += exiting ? 1.0f : 0.0f;
        wins += (exiting && pnl > 0.0f) ? 1.0f : 0.0f;
        gross_p += exiting ? fmaxf(pnl, 0.0f) : 0.0f;
        gross_l += exiting ? fmaxf(-pnl, 0.0f) : 0.0f;
        units = exiting ? 0.0f : units;
        in_pos = pos0 && !exiting;

        // entry (rare; keeps the units division off the common path —
        // the fully branchless select variant was measured SLOWER:
        // 216 vs 192 ms, the always-executed divide costs more than
        // the exec-masked branch).
        // ebit encodes t>=WARMUP && net>=entry_v; engine order: a lane
        // exiting this candle cannot re-enter the same candle (pos0).
        if (!pos0 && ebit) {
            float cost = fminf(size_pct * equity, cash);
            units = cost * fee_m / close;
            cash -= cost;
            entry_cost = cost;
            stop = close * sl_m;
            tp = close * tp_m;
            peak = close;
            trail_arm = close * act_m;
            in_pos = true;
        }

        float new_eq = cash + units * close;
        equity = new_eq;
        return new_eq;
    }
};

// ---------------------------------------------------------------------
// Kernel 1: flags. Grid = nshards x nsym x chunks blocks of 256 lanes.
// ---------------------------------------------------------------------
__global__ void __launch_bounds__(BT_BLOCK) bt_flags_kernel(
    const float* __restrict__ candles,      // (nsym, T, 4)
    const float* __restrict__ pop,          // (P, NPARAM)
    unsigned long long* __restrict__ eflags,  // (nsym, nwords, P)
    unsigned long long* __restrict__ xflags,
    int nsym, int T, int P, int chunks, int nshards, int tail,
    int shard0)                             // first shard of this launch
{
#pragma clang fp contract(off)
    __shared__ float chist[BT_SPAN];
    __shared__ double csq[BT_SPAN];   // (double)close^2, shared: saves
                                      // two f64 muls per candle per lane
    __shared__ float hl[BT_SPAN][2];
    __shared__ float4 sh_vote[BT_TILE];

    const int bid = blockIdx.x;
    const int shard = shard0 + bid / (nsym * chunks);
    const int rem = bid % (nsym * chunks);
    const int sym = rem / chunks;
    const int chunk = rem % chunks;
    const int tid = threadIdx.x;
    const int p = chunk * BT_BLOCK + tid;
    const bool act = p < P;
    const long nwords = (T + 63) >> 6;

    FlagState st;
    st.load(pop + (long)(act ? p : 0) * BT_NPARAM);

    // balanced RESNAP-aligned shard boundaries: floor(s*T/S) to the
    // RESNAP grid. A uniform body with the remainder in the last shard
    // left it 5x longer at RESNAP=16384 (T=1M, S=16) — tail-latency
    // bound the whole launch (measured 331 -> 415+ G/s fixing this).
    const int lo = (int)(((long)shard * T / nshards)
                         / BT_RESNAP * BT_RESNAP);
    const int hi = (shard == nshards - 1)
                       ? T
                       : (int)(((long)(shard + 1) * T / nshards)
                               / BT_RESNAP * BT_RESNAP);
    const int start = (shard == 0) ? 0 : lo - tail;

    const float4* sym_candles =
        reinterpret_cast<const float4*>(candles + (long)sym * T * 4);

    float prev_close = (start > 0) ? sym_candles[start - 1].x : 0.f;
    if (shard != 0) {
        // warm-tail init mirrors the t==0 branch: EMA seeded at the
        // first tail close; sig/avg_* at zero. All contract below an
        // f32 ulp within `tail` candles (see header).
        float c0 = sym_candles[start].x;
        st.ema_f = c0;
        st.ema_s = c0;
    }

    unsigned long long ew = 0ull, xw = 0ull;

    for (int t0 = start; t0 < hi; t0 += BT_TILE) {
        __syncthreads();
        for (int i = tid; i < BT_SPAN; i += BT_BLOCK) {
            const int t = t0 - BT_HALO + i;
            if (t >= 0 && t < T) {
                float4 c = sym_candles[t];
                chist[i] = c.x;
                csq[i] = (double)c.x * (double)c.x;
                hl[i][0] = c.y;
                hl[i][1] = c.z;
            } else {
                chist[i] = 0.0f;
                csq[i] = 0.0;
                hl[i][0] = 0.0f;
                hl[i][1] = 0.0f;
            }
        }
        __syncthreads();
        const int tend = min(BT_TILE, hi - t0);

        if (t0 < lo) {
            // pure warm-tail tile (tail is a multiple of BT_TILE, so
            // tiles never straddle the body boundary): recurrences only.
            // First tail candle: ema was seeded AT this close, so the
            // ema update is a bitwise no-op (a*(c-c) == 0) — mirroring
            // the t==0 init semantics without a special case.
            for (int tt = 0; tt < tend; ++tt) {
                const float close = chist[tt + BT_HALO];
                st.warm_step(close, close - prev_close);
                prev_close = close;
            }
            continue;
        }

        // body tile: shared vote series (exact, finite-window)
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

        // BB resnap at RESNAP-aligned tiles (all shards see the same
        // aligned boundaries -> bit-identical to the sequential run)
        const bool resnap_tile = (t0 > 0) && ((t0 & (BT_RESNAP - 1)) == 0);
        int tt0 = 0;
        if (resnap_tile) {
            st.resnap(&chist[BT_HALO]);
            // peeled resnap candle (compile-time SKIP_BB)
            const float close = chist[BT_HALO];
            uint2 b = st.candle<true>(
                t0, close, close - prev_close,
                chist[BT_HALO - st.bb_w], csq[BT_HALO],
                csq[BT_HALO - st.bb_w], sh_vote[0]);
            ew |= (unsigned long long)b.x << (t0 & 63);
            xw |= (unsigned long long)b.y << (t0 & 63);
            prev_close = close;
            tt0 = 1;
        }
        for (int tt = tt0; tt < tend; ++tt) {
            const int t = t0 + tt;
            const float close = chist[tt + BT_HALO];
            const float change = (t == 0) ? 0.0f : close - prev_close;
            uint2 b = st.candle<false>(
                t, close, change, chist[tt + BT_HALO - st.bb_w],
                csq[tt + BT_HALO], csq[tt + BT_HALO - st.bb_w],
                sh_vote[tt]);
            const int bit = t & 63;
            ew |= (unsigned long long)b.x << bit;
            xw |= (unsigned long long)b.y << bit;
            if (bit == 63 || t == T - 1) {
                if (act) {
                    const long w = t >> 6;
                    eflags[((long)sym * nwords + w) * P + p] = ew;
                    xflags[((long)sym * nwords + w) * P + p] = xw;
                }
                ew = 0ull;
                xw = 0ull;
            }
            prev_close = close;
        }
    }
}

// ---------------------------------------------------------------------
// Kernel 2: trades. Grid = nsym x chunks blocks of 512 threads
// (XCD-affine mapping as in backtest.hip).
//
// Producer-consumer split: waves 0-3 (tid 0-255, one lane per param)
// run the position state machine; waves 4-7 (tid 256-511, same params
// mirrored) run the mark accumulation — the two serial IEEE divisions
// (per-candle return r and drawdown) plus the sum_ret/sum_ret2/max_eq/
// max_dd chains. The equity series crosses waves through a
// double-buffered LDS tile (one 64-candle word per phase), so each
// SIMD hosts one producer and one consumer wave and the ~50% of issue
// slots the single-wave version lost to dependency stalls are filled
// by the other wave. All f32 values and op order are bit-identical to
// engine_cpu (the consumer reads the exact f32 equity values the
// producer assigned).
// ---------------------------------------------------------------------
// dynamic TradeState slots persisted between chunked launches:
// producer owns [0..12], consumer owns [13..16]
#define BT_NSTATE 18
#define BT_TBLOCK 512

__global__ void __launch_bounds__(BT_TBLOCK) bt_trades_kernel(
    const float* __restrict__ candles,       // offset to symbol group
    const float* __restrict__ pop,
    const unsigned long long* __restrict__ eflags,  // offset to group
    const unsigned long long* __restrict__ xflags,
    float* __restrict__ metrics,             // (P, nsym_stride, NMETRIC)
    int nsym, int T, int P, int chunks_per_sym, float initial_equity,
    int sym0, int nsym_stride,               // group offset + full stride
    int t_lo, int t_hi,                      // candle range [t_lo, t_hi)
    float* __restrict__ state)               // (P, nsym, NSTATE) carry
{
#pragma clang fp contract(off)
    __shared__ float sc[BT_TILE];
    __shared__ float sh[BT_TILE];
    __shared__ float sl[BT_TILE];
    // equity handoff in 32-candle half-words: 64 KB (not 128) so flags
    // blocks can co-reside on the CU during the overlap schedule
    __shared__ float eqbuf[2][32][256];
    __shared__ int skipw[2][4];              // per-producer-wave skip

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
    const bool is_prod = tid < 256;
    const int lane = tid & 255;
    const int wv = (lane >> 6);              // wave index within half
    const int p = chunk * 256 + lane;
    const bool act = p < P;
    const long nwords = (T + 63) >> 6;

    TradeState st;
    st.load(pop + (long)(act ? p : 0) * BT_NPARAM, initial_equity);
    float* my_state =
        state ? state + ((long)p * nsym + sym) * BT_NSTATE : nullptr;

    // consumer-side accumulators
    float eq_prev = initial_equity;
    float max_eq = initial_equity, max_dd = 0.f;
    float sum_ret = 0.f, sum_ret2 = 0.f;
    if (t_lo > 0 && act) {
        if (is_prod) {
            st.restore(my_state);
        } else {
            eq_prev = my_state[8];
            max_eq = my_state[13];
            max_dd = my_state[14];
            sum_ret = my_state[15];
            sum_ret2 = my_state[16];
        }
    }

    const float4* sym_candles =
        reinterpret_cast<const float4*>(candles + (long)sym * T * 4);
    const unsigned long long* esym = eflags + (long)sym * nwords * P;
    const unsigned long long* xsym = xflags + (long)sym * nwords * P;

    const int h_lo = t_lo >> 5;              // t_lo is tile-aligned
    const int h_hi = (t_hi + 31) >> 5;       // 32-candle half-words
    const int nh = h_hi - h_lo;
    unsigned long long ewrd = 0ull, xwrd = 0ull;

    for (int hi_ = 0; hi_ <= nh; ++hi_) {
        const int habs = h_lo + hi_;
        if (hi_ < nh && (habs & 7) == 0) {
            // stage the 256-candle tile this half opens (producers only
            // read sc/sh/sl; consumers only read eqbuf — no conflict)
            const int t0 = habs << 5;
            for (int i = tid; i < BT_TILE; i += BT_TBLOCK) {
                const int t = t0 + i;
                if (t < T) {
                    float4 c = sym_candles[t];
                    sc[i] = c.x;
                    sh[i] = c.y;
                    sl[i] = c.z;
                }
            }
            __syncthreads();
        }
        if (is_prod) {
            if (hi_ < nh) {
                const int buf = hi_ & 1;
                const int base = (habs & 7) << 5;
                const int hlen = min(32, t_hi - (habs << 5));
                if ((habs & 1) == 0) {
                    const long w = habs >> 1;
                    ewrd = act ? esym[w * P + p] : 0ull;
                    xwrd = act ? xsym[w * P + p] : 0ull;
                }
                const int kbit = (habs & 1) << 5;
                // exact half-word skip: flat + settled + no entry bit
                // for every lane of the wave -> the 32 candles are
                // provably no-ops (engine_cpu semantics); the consumer
                // skips via skipw (equity series constant)
                const unsigned long long ehalf =
                    (ewrd >> kbit) & 0xffffffffull;
                const bool busy = st.in_pos ||
                                  (st.cash != st.equity) || ehalf != 0ull;
                const bool skip = __ballot(busy) == 0ull;
                if ((lane & 63) == 0)
                    skipw[buf][wv] = skip ? 1 : 0;
                if (!skip) {
#pragma unroll 8
                    for (int k = 0; k < 32; ++k) {
                        if (k >= hlen) break;
                        const int i = base + k;
                        float eq = st.pstep_pos(
                            sc[i], sh[i], sl[i],
                            (ewrd >> (kbit + k)) & 1ull,
                            (xwrd >> (kbit + k)) & 1ull);
                        eqbuf[buf][k][lane] = eq;
                    }
                }
            }
        } else if (hi_ > 0) {
            const int buf = (hi_ - 1) & 1;
            if (!skipw[buf][wv]) {
                const int hprev = habs - 1;
                const int hlen = min(32, t_hi - (hprev << 5));
#pragma unroll 8
                for (int k = 0; k < 32; ++k) {
                    if (k >= hlen) break;
                    float eq = eqbuf[buf][k][lane];
                    float r = eq / eq_prev - 1.0f;
                    sum_ret += r;
                    sum_ret2 += r * r;
                    eq_prev = eq;
                    max_eq = fmaxf(max_eq, eq);
                    max_dd = fmaxf(max_dd, (max_eq - eq) / max_eq);
                }
            }
        }
        __syncthreads();
    }

    if (t_hi < T) {
        if (act) {
            if (is_prod) {
                st.save(my_state);
            } else {
                my_state[13] = max_eq;
                my_state[14] = max_dd;
                my_state[15] = sum_ret;
                my_state[16] = sum_ret2;
            }
        }
        return;
    }

    // finalize: producer publishes its five trade metrics through LDS;
    // the consumer computes sharpe/fitness and writes all ten
    // (identical formulas to backtest.hip BtState::finalize)
    if (is_prod) {
        eqbuf[0][0][lane] = st.equity;
        eqbuf[0][1][lane] = st.n_trades;
        eqbuf[0][2][lane] = st.wins;
        eqbuf[0][3][lane] = st.gross_p;
        eqbuf[0][4][lane] = st.gross_l;
    }
    __syncthreads();
    if (!is_prod && act) {
        float equity = eqbuf[0][0][lane];
        float n_trades = eqbuf[0][1][lane];
        float wins = eqbuf[0][2][lane];
        float gross_p = eqbuf[0][3][lane];
        float gross_l = eqbuf[0][4][lane];
        float n = (float)max(T, 1);
        float mean_r = sum_ret / n;
        float var_r = fmaxf(sum_ret2 / n - mean_r * mean_r, 0.0f);
        float sharpe =
            mean_r / fmaxf(sqrtf(var_r), BT_EPS) * BT_ANNUALIZE;
        if (!(n_trades > 0.0f)) sharpe = 0.0f;
        float win_rate = wins / fmaxf(n_trades, 1.0f);
        float fitness = (n_trades > 0.0f)
                            ? sharpe + win_rate - 2.0f * max_dd
                            : -1.0f;
        float* out =
            metrics + ((long)p * nsym_stride + sym0 + sym) * BT_NMETRIC;
        out[0] = equity; out[1] = n_trades; out[2] = wins;
        out[3] = gross_p; out[4] = gross_l; out[5] = max_dd;
        out[6] = sum_ret; out[7] = sum_ret2;
        out[8] = sharpe; out[9] = fitness;
    }
}

}  // namespace

// shard-group launch: shards [shard0, shard0+nlaunch) of nshards total
extern "C" void launch_bt_flags(const float* candles, const float* pop,
                                unsigned long long* eflags,
                                unsigned long long* xflags,
                                int nsym, int T, int P, int nshards,
                                int tail, int shard0, int nlaunch,
                                hipStream_t stream) {
    int chunks = (P + BT_BLOCK - 1) / BT_BLOCK;
    if (nlaunch <= 0)
        nlaunch = nshards - shard0;
    hipLaunchKernelGGL(bt_flags_kernel,
                       dim3(nlaunch * nsym * chunks), dim3(BT_BLOCK), 0,
                       stream, candles, pop, eflags, xflags, nsym, T, P,
                       chunks, nshards, tail, shard0);
}

extern "C" void launch_bt_trades(const float* candles, const float* pop,
                                 const unsigned long long* eflags,
                                 const unsigned long long* xflags,
                                 float* metrics, int nsym, int T, int P,
                                 float initial_equity, int sym0,
                                 int nsym_stride, int t_lo, int t_hi,
                                 float* state, hipStream_t stream) {
    int chunks = (P + 255) / 256;
    hipLaunchKernelGGL(bt_trades_kernel, dim3(nsym * chunks),
                       dim3(BT_TBLOCK), 0, stream, candles, pop, eflags,
                       xflags, metrics, nsym, T, P, chunks,
                       initial_equity, sym0, nsym_stride, t_lo, t_hi,
                       state);
}
