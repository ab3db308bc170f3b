"""GPU tests for the time-parallel continuous backtest
(ops/hip/backtest_tp.hip): exactness of the flags->trades decomposition
on device, warm-tail reconvergence across shard counts, and parity with
the sequential engines."""

import numpy as np
import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda:0")


def _market(T, nsym, seed=0):
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    return candles_chl_v(generate_ohlcv(T, nsym, seed=seed))


def test_continuous_matches_cpu_bitwise(dev):
    """Full pipeline (flags kernel sharded 4x + trades kernel) vs the
    sequential numpy engine: bitwise metrics. T spans 4 RESNAP periods
    so warm tails, resnap boundaries and shard stitching all fire."""
    from ai_crypto_trader_amd.backtesting.engine_cpu import run_backtest_cpu
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.ops.backtest import run_backtest_continuous_gpu

    candles = _market(49152, 2, seed=21)     # 3 RESNAP periods
    pop = random_population(32, seed=9)
    m_cpu = run_backtest_cpu(candles, pop)
    c_t = torch.from_numpy(candles).to(dev)
    p_t = torch.from_numpy(pop).to(dev)
    m_gpu = run_backtest_continuous_gpu(c_t, p_t, nshards=3).cpu().numpy()
    assert np.array_equal(m_cpu, m_gpu), (
        f"max abs diff {np.max(np.abs(m_cpu - m_gpu))}"
    )
    assert m_cpu[..., 1].sum() > 0           # trades actually happened


def test_continuous_shard_invariance(dev):
    """nshards=1 (no warm tails at all) vs nshards=4: bitwise-identical
    metrics — the empirical warm-tail reconvergence claim."""
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.ops.backtest import run_backtest_continuous_gpu

    candles = _market(49152, 2, seed=5)
    pop = random_population(64, seed=3)
    c_t = torch.from_numpy(candles).to(dev)
    p_t = torch.from_numpy(pop).to(dev)
    m1 = run_backtest_continuous_gpu(c_t, p_t, nshards=1).cpu().numpy()
    m4 = run_backtest_continuous_gpu(c_t, p_t, nshards=3).cpu().numpy()
    assert np.array_equal(m1, m4)


def test_continuous_matches_segmented_kernel_seg1(dev):
    """Continuous pipeline vs the classic one-lane-per-history kernel
    (backtest.hip) at segments=1 — two independent GPU implementations
    of the same sequential contract."""
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.ops.backtest import (
        run_backtest_continuous_gpu, run_backtest_gpu,
    )

    candles = _market(12288, 3, seed=13)
    pop = random_population(48, seed=17)
    c_t = torch.from_numpy(candles).to(dev)
    p_t = torch.from_numpy(pop).to(dev)
    m_seq = run_backtest_gpu(c_t, p_t).cpu().numpy()
    m_tp = run_backtest_continuous_gpu(c_t, p_t).cpu().numpy()
    assert np.array_equal(m_seq, m_tp)


def test_continuous_nondivisible_shapes(dev):
    """P not a multiple of 256 and T not a multiple of 64*RESNAP:
    padding lanes and the partial final flag word stay correct."""
    from ai_crypto_trader_amd.backtesting.engine_cpu import run_backtest_cpu
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.ops.backtest import run_backtest_continuous_gpu

    candles = _market(10007, 1, seed=2)      # odd T: partial final word
    pop = random_population(37, seed=4)
    m_cpu = run_backtest_cpu(candles, pop)
    c_t = torch.from_numpy(candles).to(dev)
    p_t = torch.from_numpy(pop).to(dev)
    m_gpu = run_backtest_continuous_gpu(c_t, p_t, nshards=2).cpu().numpy()
    assert np.array_equal(m_cpu, m_gpu)


def test_continuous_in_ga_engine(dev):
    """GAEngine(continuous=True) runs a generation end-to-end on the
    time-parallel path and produces the same fitness as the classic
    kernel path at segments=1."""
    from ai_crypto_trader_amd.backtesting.ga_engine import GAEngine

    candles = _market(12288, 2, seed=8)
    e1 = GAEngine(candles, pop_per_rank=64, device=dev, seed=2,
                  segments=1, continuous=False)
    e2 = GAEngine(candles, pop_per_rank=64, device=dev, seed=2,
                  segments=1, continuous=True)
    f1 = e1.eval_fitness().cpu().numpy()
    f2 = e2.eval_fitness().cpu().numpy()
    assert np.array_equal(f1, f2)


def test_continuous_time_chunked_resume(dev):
    """Chunked trades launches (bitwise state carry through HBM between
    time groups) vs the single-launch path and the CPU engine."""
    from ai_crypto_trader_amd.backtesting.engine_cpu import run_backtest_cpu
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.ops.backtest import run_backtest_continuous_gpu

    candles = _market(65536, 2, seed=31)     # 4 RESNAP periods
    pop = random_population(32, seed=12)
    m_cpu = run_backtest_cpu(candles, pop)
    c_t = torch.from_numpy(candles).to(dev)
    p_t = torch.from_numpy(pop).to(dev)
    m1 = run_backtest_continuous_gpu(c_t, p_t, nshards=4,
                                     time_groups=1).cpu().numpy()
    m3 = run_backtest_continuous_gpu(c_t, p_t, nshards=4,
                                     time_groups=3).cpu().numpy()
    m8 = run_backtest_continuous_gpu(c_t, p_t, nshards=4,
                                     time_groups=4).cpu().numpy()
    assert np.array_equal(m_cpu, m1)
    assert np.array_equal(m_cpu, m3)
    assert np.array_equal(m_cpu, m8)


def test_continuous_random_shape_fuzz(dev):
    """Property fuzz: random (T, nsym, P, nshards, time_groups, tail)
    configurations must ALL be bitwise-equal to the sequential CPU
    engine — shard alignment, partial words, padding lanes, chunked
    resume and warm tails under one sweep."""
    from ai_crypto_trader_amd.backtesting.engine_cpu import run_backtest_cpu
    from ai_crypto_trader_amd.backtesting.strategy import (
        RESNAP, random_population,
    )
    from ai_crypto_trader_amd.ops.backtest import run_backtest_continuous_gpu

    rng = np.random.default_rng(77)
    for case in range(7):
        T = int(rng.integers(2_000, 60_000))
        nsym = int(rng.integers(1, 4))
        # last case: multi-chunk population (P > 256 exercises the
        # chunk indexing in both kernels)
        P = 300 if case == 6 else int(rng.integers(3, 70))
        nshards = int(rng.integers(1, 5))
        tgroups = int(rng.integers(1, 5))
        tail = 256 * int(rng.integers(4, 9))
        candles = _market(T, nsym, seed=1000 + case)
        pop = random_population(P, seed=2000 + case)
        m_cpu = run_backtest_cpu(candles, pop)
        m_gpu = run_backtest_continuous_gpu(
            torch.from_numpy(candles).to(dev),
            torch.from_numpy(pop).to(dev),
            nshards=nshards, time_groups=tgroups, tail=tail,
        ).cpu().numpy()
        assert np.array_equal(m_cpu, m_gpu), (
            f"case {case}: T={T} nsym={nsym} P={P} nshards={nshards} "
            f"tgroups={tgroups} tail={tail} RESNAP={RESNAP} "
            f"maxdiff={np.max(np.abs(m_cpu - m_gpu))}"
        )
