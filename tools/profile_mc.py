"""Standalone MC MFMA kernel driver for rocprofv3 PMC capture.

Runs only the mc_paths_mfma dispatches (plus minimal setup) so the PMC
rows are unambiguous. Usage (on the GPU box):
    rocprofv3 --pmc SQ_INSTS_VALU,SQ_INSTS_MFMA,... -d out -- \
        python tools/profile_mc.py [--paths N] [--reps R]
"""

import argparse
import time

import numpy as np
import torch

from ai_crypto_trader_amd.ops.montecarlo import mc_paths_gpu


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--paths", type=int, default=2_000_000)
    ap.add_argument("--assets", type=int, default=64)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--no-mfma", action="store_true")
    args = ap.parse_args()

    A = args.assets
    rho = 0.4
    corr = np.full((A, A), rho) + (1 - rho) * np.eye(A)
    chol = np.linalg.cholesky(corr).astype(np.float32)
    mu = np.full(A, 0.1, np.float32)
    sigma = np.full(A, 0.5, np.float32)

    use_mfma = not args.no_mfma
    w = np.full(A, 1.0 / A, np.float32)
    kw = dict(n_steps=args.steps, n_paths=args.paths, dt=1.0 / 365.0,
              seed=7, use_mfma=use_mfma)
    mc_paths_gpu(chol, mu, sigma, w, **kw)        # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        fv, dd = mc_paths_gpu(chol, mu, sigma, w, **kw)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.reps
    print(f"paths/s {args.paths / dt:.3e}  ms {dt * 1e3:.2f} "
          f"mean {float(fv.mean()):.4f}")


if __name__ == "__main__":
    main()
