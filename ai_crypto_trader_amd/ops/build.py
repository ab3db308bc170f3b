"""In-tree hipcc build for the gfx950 kernel extension.

Builds ai_crypto_trader_amd/ops/_hip_ops.so directly with hipcc
(--offload-arch=gfx950): no JIT cache outside the tree, so the built .so
travels with repo snapshots to GPU boxes. Incremental: per-source .o files
under ops/.build keyed on mtimes.
"""

from __future__ import annotations

import subprocess
import sys
import sysconfig
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
HIP_DIR = OPS_DIR / "hip"
BUILD_DIR = OPS_DIR / ".build"
SO_PATH = OPS_DIR / "_hip_ops.so"

SOURCES = [
    "backtest.hip",
    "backtest_tp.hip",
    "ga.hip",
    "montecarlo.hip",
    "covar.hip",
    "indicators.hip",
    "lstm.hip",
    "gru.hip",
    "attention.hip",
    "laggedcorr.hip",
    "histogram.hip",
    "rl_env.hip",
    "bindings.cpp",
]

HIPCC = "hipcc"
ARCH = "gfx950"


def _include_flags() -> list[str]:
    import pybind11

    return [
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
    ]


def _compile_flags() -> list[str]:
    return [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-x",
        "hip",
    ] + _include_flags()


def build(verbose: bool = True, force: bool = False) -> Path:
    BUILD_DIR.mkdir(exist_ok=True)
    common_hpp = HIP_DIR / "common.hpp"
    objs = []
    relink = force or not SO_PATH.exists()
    for src_name in SOURCES:
        src = HIP_DIR / src_name
        if not src.exists():
            continue
        obj = BUILD_DIR / (src_name + ".o")
        objs.append(obj)
        deps_mtime = max(src.stat().st_mtime, common_hpp.stat().st_mtime)
        if not force and obj.exists() and obj.stat().st_mtime > deps_mtime:
            continue
        cmd = [HIPCC, *_compile_flags(), "-c", str(src), "-o", str(obj)]
        if verbose:
            print("[build]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
        relink = True
    if relink:
        cmd = [
            HIPCC,
            f"--offload-arch={ARCH}",
            "-shared",
            "-fPIC",
            *[str(o) for o in objs],
            "-o",
            str(SO_PATH),
        ]
        if verbose:
            print("[link]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {SO_PATH}")
