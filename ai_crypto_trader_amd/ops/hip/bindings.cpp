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

extern "C" void launch_backtest(const float*, const float*, float*, int, int,
                                int, float, hipStream_t);
extern "C" void launch_ga_evolve(const float*, const float*, const int*,
                                 const float*, float*, int, int, int, float,
                                 float, float, uint64_t, uint64_t,
                                 hipStream_t);
extern "C" void launch_mc_paths(const float*, const float*, const float*,
                                const float*, float*, float*, int, int, long,
                                float, uint64_t, hipStream_t);
extern "C" void launch_cov(const float*, float*, int, int, hipStream_t);
extern "C" void launch_indicators(const float*, float*, int, int, int,
                                  hipStream_t);
extern "C" void launch_lstm_cell_fwd(const float*, const float*, float*,
                                     float*, float*, int, int, hipStream_t);
extern "C" void launch_mfma_gemm_test(const void*, const void*, float*, int,
                                      int, int, hipStream_t);

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
             uint64_t seed, uintptr_t stream) {
              launch_mc_paths(reinterpret_cast<const float*>(chol),
                              reinterpret_cast<const float*>(drift),
                              reinterpret_cast<const float*>(vol_sqrt_dt),
                              reinterpret_cast<const float*>(weights),
                              reinterpret_cast<float*>(final_value),
                              reinterpret_cast<float*>(max_dd), n_assets,
                              n_steps, n_paths, s0, seed, as_stream(stream));
              check(hipGetLastError(), "mc_paths launch");
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

    m.def("device_synchronize", []() { check(hipDeviceSynchronize(), "sync"); });
}
