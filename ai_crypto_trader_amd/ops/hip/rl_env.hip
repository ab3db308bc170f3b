// GPU-resident trading environments + GAE reduction.
//
// Replaces reinforcement_learning.py's gym-style python env stepping
// (:421-503) with one CDNA4 lane per environment: the env state machine
// (indicator recurrences + position + equity, the same per-candle math as
// the backtest kernel) advances all E envs one candle per launch, entirely
// on-device — the policy net (PyTorch-ROCm) and the env alternate on one
// HIP stream with no host round-trip. 256+ envs/GPU, episodes are windows
// of the synthetic candle tensor, reset offsets drawn from Philox.
//
// gae_kernel: per-env reverse scan for GAE(lambda) advantages
// (the PPO advantage reduction the north star names): one lane per env,
// serial over T in registers — A_t = delta_t + gamma*lambda*A_{t+1}.

#include "common.hpp"

#define ENV_NSTATE 16
#define ENV_NOBS 12
#define ENV_NACT 3

// state slots
#define S_SYM 0
#define S_T 1
#define S_EMA_F 2
#define S_EMA_S 3
#define S_SIG 4
#define S_AVG_GAIN 5
#define S_AVG_LOSS 6
#define S_PREV_CLOSE 7
#define S_CASH 8
#define S_UNITS 9
#define S_ENTRY 10
#define S_EQUITY 11
#define S_EP_LEFT 12
#define S_PREV_EQ 13

namespace {

__device__ void write_obs(float* obs, const float* st, float rsi,
                          float macd_hist, float r1, float r5, float r15,
                          float close) {
    const float units = st[S_UNITS];
    const float equity = st[S_EQUITY];
    const float in_pos = units > 0.0f ? 1.0f : 0.0f;
    const float upnl =
        in_pos > 0.0f ? (close / st[S_ENTRY] - 1.0f) : 0.0f;
    obs[0] = rsi * 0.01f;
    obs[1] = macd_hist / close * 100.0f;
    obs[2] = r1 * 100.0f;
    obs[3] = r5 * 100.0f;
    obs[4] = r15 * 100.0f;
    obs[5] = in_pos;
    obs[6] = upnl * 10.0f;
    obs[7] = equity - 1.0f;
    obs[8] = st[S_AVG_GAIN] * 1000.0f;
    obs[9] = st[S_AVG_LOSS] * 1000.0f;
    obs[10] = (st[S_EMA_F] / st[S_EMA_S] - 1.0f) * 100.0f;
    obs[11] = st[S_EP_LEFT] * 0.001f;
}

__global__ void env_reset_kernel(const float* __restrict__ candles,
                                 float* __restrict__ state,
                                 float* __restrict__ obs, int nsym, int T,
                                 int n_envs, int ep_len, uint64_t seed,
                                 uint64_t epoch) {
    int e = blockIdx.x * blockDim.x + threadIdx.x;
    if (e >= n_envs) return;
    Philox4 r = philox4x32(seed, (uint64_t)e, epoch);
    int sym = (int)(r.x % (uint32_t)nsym);
    int max_start = T - ep_len - 32;
    int t0 = 16 + (int)(r.y % (uint32_t)(max_start > 1 ? max_start : 1));
    float* st = state + (long)e * ENV_NSTATE;
    const float c0 = candles[((long)sym * T + t0) * 4];
    st[S_SYM] = (float)sym;
    st[S_T] = (float)t0;
    st[S_EMA_F] = c0; st[S_EMA_S] = c0; st[S_SIG] = 0.0f;
    st[S_AVG_GAIN] = 0.0f; st[S_AVG_LOSS] = 0.0f;
    st[S_PREV_CLOSE] = c0;
    st[S_CASH] = 1.0f; st[S_UNITS] = 0.0f; st[S_ENTRY] = c0;
    st[S_EQUITY] = 1.0f; st[S_PREV_EQ] = 1.0f;
    st[S_EP_LEFT] = (float)ep_len;
    write_obs(obs + (long)e * ENV_NOBS, st, 50.0f, 0.0f, 0.0f, 0.0f, 0.0f,
              c0);
}

__global__ void env_step_kernel(const float* __restrict__ candles,
                                float* __restrict__ state,
                                const int* __restrict__ actions,
                                float* __restrict__ obs,
                                float* __restrict__ reward,
                                float* __restrict__ done, int nsym, int T,
                                int n_envs, int ep_len, float fee,
                                uint64_t seed, uint64_t epoch) {
    int e = blockIdx.x * blockDim.x + threadIdx.x;
    if (e >= n_envs) return;
    float* st = state + (long)e * ENV_NSTATE;
    const int sym = (int)st[S_SYM];
    int t = (int)st[S_T] + 1;
    const float4* sc =
        reinterpret_cast<const float4*>(candles + (long)sym * T * 4);
    const float4 c4 = sc[t];
    const float close = c4.x;

    // indicators (same recurrences as backtest.hip, fixed standard periods)
    const float a_f = 2.0f / 13.0f, a_s = 2.0f / 27.0f, a_sig = 2.0f / 10.0f;
    float ema_f = st[S_EMA_F] + a_f * (close - st[S_EMA_F]);
    float ema_s = st[S_EMA_S] + a_s * (close - st[S_EMA_S]);
    const float macd = ema_f - ema_s;
    float sig = st[S_SIG] + a_sig * (macd - st[S_SIG]);
    const float change = close - st[S_PREV_CLOSE];
    float ag = st[S_AVG_GAIN] + (fmaxf(change, 0.f) - st[S_AVG_GAIN]) / 14.f;
    float al = st[S_AVG_LOSS] + (fmaxf(-change, 0.f) - st[S_AVG_LOSS]) / 14.f;
    const float rsi = 100.0f - 100.0f / (1.0f + ag / fmaxf(al, 1e-9f));

    // action: 0 hold, 1 enter long (all-in), 2 close position
    const int act = actions[e];
    float cash = st[S_CASH], units = st[S_UNITS], entry = st[S_ENTRY];
    if (act == 1 && units == 0.0f) {
        units = cash * (1.0f - fee) / close;
        entry = close;
        cash = 0.0f;
    } else if (act == 2 && units > 0.0f) {
        cash += units * close * (1.0f - fee);
        units = 0.0f;
    }
    const float equity = cash + units * close;
    const float prev_eq = st[S_EQUITY];
    const float rwd = __logf(fmaxf(equity, 1e-9f) / fmaxf(prev_eq, 1e-9f));

    float ep_left = st[S_EP_LEFT] - 1.0f;
    const bool ep_done = ep_left <= 0.0f || t + 2 >= T;

    const float r1 = close / st[S_PREV_CLOSE] - 1.0f;
    const float c5 = sc[max(t - 5, 0)].x;
    const float c15 = sc[max(t - 15, 0)].x;

    st[S_EMA_F] = ema_f; st[S_EMA_S] = ema_s; st[S_SIG] = sig;
    st[S_AVG_GAIN] = ag; st[S_AVG_LOSS] = al;
    st[S_PREV_CLOSE] = close;
    st[S_CASH] = cash; st[S_UNITS] = units; st[S_ENTRY] = entry;
    st[S_EQUITY] = equity; st[S_PREV_EQ] = prev_eq;
    st[S_T] = (float)t;
    st[S_EP_LEFT] = ep_left;

    reward[e] = rwd;
    done[e] = ep_done ? 1.0f : 0.0f;
    write_obs(obs + (long)e * ENV_NOBS, st, rsi, macd - sig, r1,
              close / c5 - 1.0f, close / c15 - 1.0f, close);

    if (ep_done) {
        // in-place reset (fresh episode, epoch-salted Philox)
        Philox4 r = philox4x32(seed, (uint64_t)e,
                               0x100000000ull * epoch + (uint64_t)t);
        int nsym_ = nsym;
        int sym2 = (int)(r.x % (uint32_t)nsym_);
        int max_start = T - ep_len - 32;
        int t0 = 16 + (int)(r.y % (uint32_t)(max_start > 1 ? max_start : 1));
        const float c0 = candles[((long)sym2 * T + t0) * 4];
        st[S_SYM] = (float)sym2;
        st[S_T] = (float)t0;
        st[S_EMA_F] = c0; st[S_EMA_S] = c0; st[S_SIG] = 0.0f;
        st[S_AVG_GAIN] = 0.0f; st[S_AVG_LOSS] = 0.0f;
        st[S_PREV_CLOSE] = c0;
        st[S_CASH] = 1.0f; st[S_UNITS] = 0.0f; st[S_ENTRY] = c0;
        st[S_EQUITY] = 1.0f; st[S_PREV_EQ] = 1.0f;
        st[S_EP_LEFT] = (float)ep_len;
    }
}

// GAE(lambda): one lane per env, reverse scan over T stored steps.
// rewards/values/dones: (T, E) f32; values has T+1 rows (bootstrap).
__global__ void gae_kernel(const float* __restrict__ rewards,
                           const float* __restrict__ values,
                           const float* __restrict__ dones,
                           float* __restrict__ adv,
                           float* __restrict__ returns, int T, int E,
                           float gamma, float lam) {
    int e = blockIdx.x * blockDim.x + threadIdx.x;
    if (e >= E) return;
    float a = 0.0f;
    for (int t = T - 1; t >= 0; --t) {
        const float nonterm = 1.0f - dones[t * E + e];
        const float delta = rewards[t * E + e] +
                            gamma * values[(t + 1) * E + e] * nonterm -
                            values[t * E + e];
        a = delta + gamma * lam * nonterm * a;
        adv[t * E + e] = a;
        returns[t * E + e] = a + values[t * E + e];
    }
}

}  // namespace

extern "C" void launch_env_reset(const float* candles, float* state,
                                 float* obs, int nsym, int T, int n_envs,
                                 int ep_len, uint64_t seed, uint64_t epoch,
                                 hipStream_t stream) {
    dim3 grid((n_envs + 255) / 256);
    hipLaunchKernelGGL(env_reset_kernel, grid, dim3(256), 0, stream, candles,
                       state, obs, nsym, T, n_envs, ep_len, seed, epoch);
}

extern "C" void launch_env_step(const float* candles, float* state,
                                const int* actions, float* obs,
                                float* reward, float* done, int nsym, int T,
                                int n_envs, int ep_len, float fee,
                                uint64_t seed, uint64_t epoch,
                                hipStream_t stream) {
    dim3 grid((n_envs + 255) / 256);
    hipLaunchKernelGGL(env_step_kernel, grid, dim3(256), 0, stream, candles,
                       state, actions, obs, reward, done, nsym, T, n_envs,
                       ep_len, fee, seed, epoch);
}

extern "C" void launch_gae(const float* rewards, const float* values,
                           const float* dones, float* adv, float* returns,
                           int T, int E, float gamma, float lam,
                           hipStream_t stream) {
    dim3 grid((E + 255) / 256);
    hipLaunchKernelGGL(gae_kernel, grid, dim3(256), 0, stream, rewards,
                       values, dones, adv, returns, T, E, gamma, lam);
}
