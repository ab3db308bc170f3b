// pybind11 bindings for the gfx950 kernels. Deliberately torch-header-free:
// callers pass torch tensors' data_ptr() as integers plus the current
// torch HIP stream handle (torch.cuda.current_stream().cuda_stream), so
// this module compiles with plain hipcc and all allocation/stream
// management stays on the PyTorch-ROCm side.

#include <pybind11/pybind11.h>

#include <cstdint>
#include <hip/hip_runtime.h>
#include <stdexcept>
#include <string>

namespace py = pybind11;

extern "C" void launch_bt_flags(const float*, const float*,
                                unsigned long long*, unsigned long long*,
                                int, int, int, int, int, int, int,
                                hipStream_t);
extern "C" void launch_bt_trades(const float*, const float*,
                                 const unsigned long long*,
                                 const unsigned long long*, float*, int,
                                 int, int, float, int, int, int, int,
                                 float*, hipStream_t);
extern "C" void launch_backtest(const float*, const float*, float*, int, int,
                                int, float, hipStream_t);
extern "C" void launch_ga_evolve(const float*, const float*, const int*,
                                 const float*, float*, int, int, int, float,
                                 float, float, uint64_t, uint64_t,
                                 hipStream_t);
extern "C" void launch_mc_paths(const float*, const float*, const float*,
                                const float*, float*, float*, int, int, long,
                                float, uint64_t, long, int, hipStream_t);
extern "C" void launch_mc_paths_mfma(const float*, const float*,
                                     const float*, float*, float*, int, int,
                                     long, float, uint64_t, long, int,
                                     hipStream_t);
extern "C" void launch_mc_bootstrap(const float*, const float*, float*,
                                    float*, int, int, int, long, float,
                                    uint64_t, long, hipStream_t);
extern "C" void launch_cov(const float*, float*, int, int, hipStream_t);
extern "C" void launch_indicators(const float*, float*, int, int, int,
                                  hipStream_t);
extern "C" void launch_lstm_seq_fwd(const void*, const void*, const float*,
                                    void*, void*, float*, int, int, int, int,
                                    hipStream_t);
extern "C" void launch_lstm_seq_bwd(const float*, const void*, const float*,
                                    const void*, void*, int, int, int,
                                    hipStream_t);
extern "C" void launch_mfma_gemm_test(const void*, const void*, float*, int,
                                      int, int, hipStream_t);
extern "C" void launch_mfma_gemm_test_bf16(const void*, const void*, float*,
                                           int, int, int, hipStream_t);
extern "C" void launch_gru_seq_fwd(const void*, const void*, const float*,
                                   void*, void*, float*, int, int, int,
                                   int, hipStream_t);
extern "C" void launch_gru_seq_bwd(const float*, const void*, const float*,
                                   const void*, const void*, void*, int, int,
                                   int, hipStream_t);
extern "C" void launch_attn_fwd(const void*, const void*, const void*,
                                void*, void*, long, int, int, float,
                                hipStream_t);
extern "C" void launch_attn_bwd(const void*, const void*, const void*,
                                const void*, const void*, void*, void*,
                                void*, long, int, int, float, hipStream_t);
extern "C" void launch_lagged_corr(const float*, const float*, float*, int,
                                   int, hipStream_t);
extern "C" void launch_vp_hist(const float*, const float*, const float*,
                               float*, float*, int, int, int, hipStream_t);
extern "C" void launch_env_reset(const float*, float*, float*, int, int, int,
                                 int, uint64_t, uint64_t, hipStream_t);
extern "C" void launch_env_step(const float*, float*, const int*, float*,
                                float*, float*, int, int, int, int, float,
                                uint64_t, uint64_t, hipStream_t);
extern "C" void launch_gae(const float*, const float*, const float*, float*,
                           float*, int, int, float, float, hipStream_t);

static void check(hipError_t e, const char* what) {
    if (e != hipSuccess)
        throw std::runtime_error(std::string(what) + ": " +
                                 hipGetErrorString(e));
}

static hipStream_t as_stream(uintptr_t s) {
    return reinterpret_cast<hipStream_t>(s);
}

PYBIND11_MODULE(_hip_ops, m) {
    m.doc() = "MI355X (gfx950) kernels for ai_crypto_trader_amd";
    m.attr("GFX_ARCH") = "gfx950";

    m.def("backtest",
          [](uintptr_t candles, uintptr_t pop, uintptr_t metrics, int nsym,
             int T, int P, float initial_equity, uintptr_t stream) {
              launch_backtest(reinterpret_cast<const float*>(candles),
                              reinterpret_cast<const float*>(pop),
                              reinterpret_cast<float*>(metrics), nsym, T, P,
                              initial_equity, as_stream(stream));
              check(hipGetLastError(), "backtest launch");
          },
          py::arg("candles"), py::arg("pop"), py::arg("metrics"),
          py::arg("nsym"), py::arg("T"), py::arg("P"),
          py::arg("initial_equity"), py::arg("stream"));

    m.def("bt_flags",
          [](uintptr_t candles, uintptr_t pop, uintptr_t eflags,
             uintptr_t xflags, int nsym, int T, int P, int nshards,
             int tail, int shard0, int nlaunch, uintptr_t stream) {
              launch_bt_flags(
                  reinterpret_cast<const float*>(candles),
                  reinterpret_cast<const float*>(pop),
                  reinterpret_cast<unsigned long long*>(eflags),
                  reinterpret_cast<unsigned long long*>(xflags), nsym, T,
                  P, nshards, tail, shard0, nlaunch, as_stream(stream));
              check(hipGetLastError(), "bt_flags launch");
          },
          py::arg("candles"), py::arg("pop"), py::arg("eflags"),
          py::arg("xflags"), py::arg("nsym"), py::arg("T"), py::arg("P"),
          py::arg("nshards"), py::arg("tail"), py::arg("shard0") = 0,
          py::arg("nlaunch") = 0, py::arg("stream") = 0);

    m.def("bt_trades",
          [](uintptr_t candles, uintptr_t pop, uintptr_t eflags,
             uintptr_t xflags, uintptr_t metrics, int nsym, int T, int P,
             float initial_equity, int sym0, int nsym_stride, int t_lo,
             int t_hi, uintptr_t state, uintptr_t stream) {
              launch_bt_trades(
                  reinterpret_cast<const float*>(candles),
                  reinterpret_cast<const float*>(pop),
                  reinterpret_cast<const unsigned long long*>(eflags),
                  reinterpret_cast<const unsigned long long*>(xflags),
                  reinterpret_cast<float*>(metrics), nsym, T, P,
                  initial_equity, sym0, nsym_stride, t_lo, t_hi,
                  reinterpret_cast<float*>(state), as_stream(stream));
              check(hipGetLastError(), "bt_trades launch");
          },
          py::arg("candles"), py::arg("pop"), py::arg("eflags"),
          py::arg("xflags"), py::arg("metrics"), py::arg("nsym"),
          py::arg("T"), py::arg("P"), py::arg("initial_equity"),
          py::arg("sym0"), py::arg("nsym_stride"), py::arg("t_lo"),
          py::arg("t_hi"), py::arg("state"), py::arg("stream"));

    m.def("ga_evolve",
          [](uintptr_t pop, uintptr_t fitness, uintptr_t order,
             uintptr_t bounds, uintptr_t out, int P, int elite_k,
             int tournament, float cx_rate, float mut_rate, float mut_scale,
             uint64_t seed, uint64_t gen, uintptr_t stream) {
              launch_ga_evolve(reinterpret_cast<const float*>(pop),
                               reinterpret_cast<const float*>(fitness),
                               reinterpret_cast<const int*>(order),
                               reinterpret_cast<const float*>(bounds),
                               reinterpret_cast<float*>(out), P, elite_k,
                               tournament, cx_rate, mut_rate, mut_scale, seed,
                               gen, as_stream(stream));
              check(hipGetLastError(), "ga_evolve launch");
          });

    m.def("mc_paths",
          [](uintptr_t chol, uintptr_t drift, uintptr_t vol_sqrt_dt,
             uintptr_t weights, uintptr_t final_value, uintptr_t max_dd,
             int n_assets, int n_steps, long n_paths, float s0,
             uint64_t seed, long path_base, int antithetic,
             uintptr_t stream) {
              launch_mc_paths(reinterpret_cast<const float*>(chol),
                              reinterpret_cast<const float*>(drift),
                              reinterpret_cast<const float*>(vol_sqrt_dt),
                              reinterpret_cast<const float*>(weights),
                              reinterpret_cast<float*>(final_value),
                              reinterpret_cast<float*>(max_dd), n_assets,
                              n_steps, n_paths, s0, seed, path_base,
                              antithetic, as_stream(stream));
              check(hipGetLastError(), "mc_paths launch");
          });

    m.def("mc_paths_mfma",
          [](uintptr_t chol, uintptr_t drift, uintptr_t vol_sqrt_dt,
             uintptr_t final_value, uintptr_t max_dd, int n_assets,
             int n_steps, long n_paths, float s0, uint64_t seed,
             long path_base, int antithetic, uintptr_t stream) {
              launch_mc_paths_mfma(reinterpret_cast<const float*>(chol),
                                   reinterpret_cast<const float*>(drift),
                                   reinterpret_cast<const float*>(vol_sqrt_dt),
                                   reinterpret_cast<float*>(final_value),
                                   reinterpret_cast<float*>(max_dd),
                                   n_assets, n_steps, n_paths, s0, seed,
                                   path_base, antithetic, as_stream(stream));
              check(hipGetLastError(), "mc_paths_mfma launch");
          });

    m.def("mc_bootstrap",
          [](uintptr_t logret, uintptr_t wS0, uintptr_t final_value,
             uintptr_t max_dd, int n_assets, int t_hist, int n_steps,
             long n_paths, float v0, uint64_t seed, long path_base,
             uintptr_t stream) {
              launch_mc_bootstrap(reinterpret_cast<const float*>(logret),
                                  reinterpret_cast<const float*>(wS0),
                                  reinterpret_cast<float*>(final_value),
                                  reinterpret_cast<float*>(max_dd),
                                  n_assets, t_hist, n_steps, n_paths, v0,
                                  seed, path_base, as_stream(stream));
              check(hipGetLastError(), "mc_bootstrap launch");
          });

    m.def("cov",
          [](uintptr_t returns, uintptr_t cov, int T, int N,
             uintptr_t stream) {
              launch_cov(reinterpret_cast<const float*>(returns),
                         reinterpret_cast<float*>(cov), T, N,
                         as_stream(stream));
              check(hipGetLastError(), "cov launch");
          });

    m.def("indicators",
          [](uintptr_t candles, uintptr_t out, int nsym, int T, int nind,
             uintptr_t stream) {
              launch_indicators(reinterpret_cast<const float*>(candles),
                                reinterpret_cast<float*>(out), nsym, T, nind,
                                as_stream(stream));
              check(hipGetLastError(), "indicators launch");
          });

    m.def("mfma_gemm_test",
          [](uintptr_t a, uintptr_t b, uintptr_t c, int M, int N, int K,
             uintptr_t stream) {
              launch_mfma_gemm_test(reinterpret_cast<const void*>(a),
                                    reinterpret_cast<const void*>(b),
                                    reinterpret_cast<float*>(c), M, N, K,
                                    as_stream(stream));
              check(hipGetLastError(), "mfma_gemm_test launch");
          });

    m.def("mfma_gemm_test_bf16",
          [](uintptr_t a, uintptr_t b, uintptr_t c, int M, int N, int K,
             uintptr_t stream) {
              launch_mfma_gemm_test_bf16(reinterpret_cast<const void*>(a),
                                         reinterpret_cast<const void*>(b),
                                         reinterpret_cast<float*>(c), M, N,
                                         K, as_stream(stream));
              check(hipGetLastError(), "mfma_gemm_test_bf16 launch");
          });

    m.def("lstm_seq_fwd",
          [](uintptr_t xproj, uintptr_t Wt, uintptr_t bias, uintptr_t h_out,
             uintptr_t gates_out, uintptr_t c_out, int B, int T, int H,
             int save_mode, uintptr_t stream) {
              launch_lstm_seq_fwd(reinterpret_cast<const void*>(xproj),
                                  reinterpret_cast<const void*>(Wt),
                                  reinterpret_cast<const float*>(bias),
                                  reinterpret_cast<void*>(h_out),
                                  reinterpret_cast<void*>(gates_out),
                                  reinterpret_cast<float*>(c_out), B, T, H,
                                  save_mode, as_stream(stream));
              check(hipGetLastError(), "lstm_seq_fwd launch");
          });

    m.def("lstm_seq_bwd",
          [](uintptr_t dh_up, uintptr_t gates, uintptr_t c_sav, uintptr_t W,
             uintptr_t dgates_out, int B, int T, int H, uintptr_t stream) {
              launch_lstm_seq_bwd(reinterpret_cast<const float*>(dh_up),
                                  reinterpret_cast<const void*>(gates),
                                  reinterpret_cast<const float*>(c_sav),
                                  reinterpret_cast<const void*>(W),
                                  reinterpret_cast<void*>(dgates_out), B, T,
                                  H, as_stream(stream));
              check(hipGetLastError(), "lstm_seq_bwd launch");
          });

    m.def("env_reset",
          [](uintptr_t candles, uintptr_t state, uintptr_t obs, int nsym,
             int T, int n_envs, int ep_len, uint64_t seed, uint64_t epoch,
             uintptr_t stream) {
              launch_env_reset(reinterpret_cast<const float*>(candles),
                               reinterpret_cast<float*>(state),
                               reinterpret_cast<float*>(obs), nsym, T,
                               n_envs, ep_len, seed, epoch,
                               as_stream(stream));
              check(hipGetLastError(), "env_reset launch");
          });

    m.def("env_step",
          [](uintptr_t candles, uintptr_t state, uintptr_t actions,
             uintptr_t obs, uintptr_t reward, uintptr_t done, int nsym,
             int T, int n_envs, int ep_len, float fee, uint64_t seed,
             uint64_t epoch, uintptr_t stream) {
              launch_env_step(reinterpret_cast<const float*>(candles),
                              reinterpret_cast<float*>(state),
                              reinterpret_cast<const int*>(actions),
                              reinterpret_cast<float*>(obs),
                              reinterpret_cast<float*>(reward),
                              reinterpret_cast<float*>(done), nsym, T,
                              n_envs, ep_len, fee, seed, epoch,
                              as_stream(stream));
              check(hipGetLastError(), "env_step launch");
          });

    m.def("gae",
          [](uintptr_t rewards, uintptr_t values, uintptr_t dones,
             uintptr_t adv, uintptr_t returns, int T, int E, float gamma,
             float lam, uintptr_t stream) {
              launch_gae(reinterpret_cast<const float*>(rewards),
                         reinterpret_cast<const float*>(values),
                         reinterpret_cast<const float*>(dones),
                         reinterpret_cast<float*>(adv),
                         reinterpret_cast<float*>(returns), T, E, gamma, lam,
                         as_stream(stream));
              check(hipGetLastError(), "gae launch");
          });

    m.def("gru_seq_fwd",
          [](uintptr_t xproj, uintptr_t Wt, uintptr_t bias, uintptr_t h_out,
             uintptr_t gates_out, uintptr_t hpn_out, int B, int T, int H,
             int save_mode, uintptr_t stream) {
              launch_gru_seq_fwd(reinterpret_cast<const void*>(xproj),
                                 reinterpret_cast<const void*>(Wt),
                                 reinterpret_cast<const float*>(bias),
                                 reinterpret_cast<void*>(h_out),
                                 reinterpret_cast<void*>(gates_out),
                                 reinterpret_cast<float*>(hpn_out), B, T, H,
                                 save_mode, as_stream(stream));
              check(hipGetLastError(), "gru_seq_fwd launch");
          });

    m.def("gru_seq_bwd",
          [](uintptr_t dh_up, uintptr_t gates, uintptr_t hpn,
             uintptr_t h_out, uintptr_t W, uintptr_t dgates_out, int B,
             int T, int H, uintptr_t stream) {
              launch_gru_seq_bwd(reinterpret_cast<const float*>(dh_up),
                                 reinterpret_cast<const void*>(gates),
                                 reinterpret_cast<const float*>(hpn),
                                 reinterpret_cast<const void*>(h_out),
                                 reinterpret_cast<const void*>(W),
                                 reinterpret_cast<void*>(dgates_out), B, T,
                                 H, as_stream(stream));
              check(hipGetLastError(), "gru_seq_bwd launch");
          });

    m.def("attn_fwd",
          [](uintptr_t Q, uintptr_t K, uintptr_t V, uintptr_t O,
             uintptr_t P_out, long bh_count, int s_len, int d_head,
             float scale, uintptr_t stream) {
              launch_attn_fwd(reinterpret_cast<const void*>(Q),
                              reinterpret_cast<const void*>(K),
                              reinterpret_cast<const void*>(V),
                              reinterpret_cast<void*>(O),
                              reinterpret_cast<void*>(P_out), bh_count,
                              s_len, d_head, scale, as_stream(stream));
              check(hipGetLastError(), "attn_fwd launch");
          });

    m.def("attn_bwd",
          [](uintptr_t Q, uintptr_t K, uintptr_t V, uintptr_t P,
             uintptr_t dO, uintptr_t dQ, uintptr_t dK, uintptr_t dV,
             long bh_count, int s_len, int d_head, float scale,
             uintptr_t stream) {
              launch_attn_bwd(reinterpret_cast<const void*>(Q),
                              reinterpret_cast<const void*>(K),
                              reinterpret_cast<const void*>(V),
                              reinterpret_cast<const void*>(P),
                              reinterpret_cast<const void*>(dO),
                              reinterpret_cast<void*>(dQ),
                              reinterpret_cast<void*>(dK),
                              reinterpret_cast<void*>(dV), bh_count,
                              s_len, d_head, scale, as_stream(stream));
              check(hipGetLastError(), "attn_bwd launch");
          });

    m.def("lagged_corr",
          [](uintptr_t a, uintptr_t b, uintptr_t out, int n, int max_lag,
             uintptr_t stream) {
              launch_lagged_corr(reinterpret_cast<const float*>(a),
                                 reinterpret_cast<const float*>(b),
                                 reinterpret_cast<float*>(out), n, max_lag,
                                 as_stream(stream));
              check(hipGetLastError(), "lagged_corr launch");
          });

    m.def("vp_hist",
          [](uintptr_t candles, uintptr_t lo, uintptr_t hi, uintptr_t hist,
             uintptr_t updown, int nsym, int T, int n_bins,
             uintptr_t stream) {
              launch_vp_hist(reinterpret_cast<const float*>(candles),
                             reinterpret_cast<const float*>(lo),
                             reinterpret_cast<const float*>(hi),
                             reinterpret_cast<float*>(hist),
                             reinterpret_cast<float*>(updown), nsym, T,
                             n_bins, as_stream(stream));
              check(hipGetLastError(), "vp_hist launch");
          });

    m.def("device_synchronize", []() { check(hipDeviceSynchronize(), "sync"); });
}
