"""GPU backtest op wrapper (HIP kernel: ops/hip/backtest.hip).

CPU golden reference: backtesting/engine_cpu.run_backtest_cpu — both
implement the per-candle state machine specified in
backtesting/strategy.py.
"""

from __future__ import annotations

import torch

from . import require_hip_ops
from ..backtesting.engine_cpu import NMETRIC
from ..backtesting.strategy import NPARAM


def run_backtest_gpu(
    candles: torch.Tensor,     # (nsym, T, 4) f32 cuda
    population: torch.Tensor,  # (P, NPARAM) f32 cuda
    *,
    initial_equity: float = 1.0,
) -> torch.Tensor:             # (P, nsym, NMETRIC) f32 cuda
    ops = require_hip_ops()
    assert candles.is_cuda and population.is_cuda
    assert candles.dtype == torch.float32
    assert population.dtype == torch.float32
    assert candles.dim() == 3 and candles.shape[2] == 4
    assert population.dim() == 2 and population.shape[1] == NPARAM
    candles = candles.contiguous()
    population = population.contiguous()
    nsym, T, _ = candles.shape
    P = population.shape[0]
    metrics = torch.empty(
        (P, nsym, NMETRIC), dtype=torch.float32, device=candles.device
    )
    stream = torch.cuda.current_stream(candles.device).cuda_stream
    ops.backtest(
        candles.data_ptr(), population.data_ptr(), metrics.data_ptr(),
        nsym, T, P, float(initial_equity), stream,
    )
    return metrics


# Flag-bitmap buffers reused across GA generations (16.4 GB at the
# flagship 1024x64x1M shape — allocated once, not per step).
_flag_cache: dict = {}


def pick_nshards(nsym: int, T: int, P: int, *, target_blocks: int = 6144,
                 tail: int = 2048) -> int:
    """Time shards for bt_flags: enough blocks to oversubscribe every CU,
    bounded so each shard body is >= one RESNAP period and >= 4x the
    warm tail. 24 shards x 8 even time groups measured best end-to-end
    at the flagship shape (234 G/s; full sweep in
    profiles/backtest_continuous_pmc.json)."""
    from ..backtesting.strategy import RESNAP

    chunks = (P + 255) // 256
    base = nsym * chunks
    want = max(1, -(-target_blocks // base))
    max_shards = max(1, T // max(RESNAP, 4 * tail))
    return min(want, max_shards, 64)


_pipe_streams: dict = {}


def run_backtest_continuous_gpu(
    candles: torch.Tensor,     # (nsym, T, 4) f32 cuda
    population: torch.Tensor,  # (P, NPARAM) f32 cuda
    *,
    initial_equity: float = 1.0,
    nshards: int = 0,          # 0 = auto
    tail: int = 2048,
    time_groups: int = 0,      # 0 = auto; 1 = serial two-launch path
) -> torch.Tensor:             # (P, nsym, NMETRIC) f32 cuda
    """Continuous (unsegmented) backtest via the time-parallel kernel
    pair (ops/hip/backtest_tp.hip): indicator/vote flags computed with
    the time axis split across shards (warm-tail reconverged EMA/RSI +
    RESNAP-exact Bollinger), then the exact sequential position state
    machine over packed flag words. Same metrics contract as
    run_backtest_gpu / engine_cpu.run_backtest_cpu.

    Overlap: the flags shards launch in `time_groups` groups on stream
    F; the trades kernel is resumable over candle ranges (bitwise state
    carry in HBM) and chunk g launches on stream T as soon as flags
    group g completes. The latency-bound trades chain (1 wave/SIMD,
    ~50% issue-idle) thus walks candles co-resident with the
    issue-bound flags waves instead of after them — the chip runs both
    at once and total time approaches max(flags, trades) + first-group
    latency. (Symbol-group pipelining was measured and rejected: the
    trades wall time is the length of the serial candle chain,
    independent of lane count, so symbol groups serialize G full-length
    chains.)"""
    ops = require_hip_ops()
    assert candles.is_cuda and population.is_cuda
    assert candles.dtype == torch.float32
    assert population.dtype == torch.float32
    assert tail % 256 == 0
    candles = candles.contiguous()
    population = population.contiguous()
    nsym, T, _ = candles.shape
    P = population.shape[0]
    from ..backtesting.strategy import RESNAP

    if nshards <= 0:
        nshards = pick_nshards(nsym, T, P, tail=tail)
    # shard bodies must be RESNAP-aligned (BB restart exactness)
    nshards = min(nshards, max(1, T // RESNAP))
    nwords = (T + 63) // 64
    dev = candles.device
    key = (nsym, nwords, P, dev.index)
    bufs = _flag_cache.get(key)
    if bufs is None or bufs[0].device != dev:
        eflags = torch.empty((nsym, nwords, P), dtype=torch.int64,
                             device=dev)
        xflags = torch.empty_like(eflags)
        carry = torch.empty((P, nsym, 18), dtype=torch.float32,
                            device=dev)
        _flag_cache.clear()      # one shape live at a time (16 GB-class)
        _flag_cache[key] = (eflags, xflags, carry)
    else:
        eflags, xflags, carry = bufs
    metrics = torch.empty((P, nsym, NMETRIC), dtype=torch.float32,
                          device=dev)
    cptr = candles.data_ptr()
    pptr = population.data_ptr()

    if time_groups <= 0:
        time_groups = 8 if nshards >= 8 else 1
    time_groups = min(time_groups, nshards)
    if time_groups == 1:
        stream = torch.cuda.current_stream(dev).cuda_stream
        ops.bt_flags(cptr, pptr, eflags.data_ptr(), xflags.data_ptr(),
                     nsym, T, P, nshards, tail, 0, nshards, stream)
        ops.bt_trades(cptr, pptr, eflags.data_ptr(), xflags.data_ptr(),
                      metrics.data_ptr(), nsym, T, P,
                      float(initial_equity), 0, nsym, 0, T, 0, stream)
        return metrics

    ss = _pipe_streams.get(dev.index)
    if ss is None:
        # trades on the high-priority stream (measured neutral on
        # MI355X — CU arbitration ignores stream priority — but it
        # documents intent and costs nothing)
        ss = (torch.cuda.Stream(dev),
              torch.cuda.Stream(dev, priority=-1))
        _pipe_streams[dev.index] = ss
    sf, st = ss
    cur = torch.cuda.current_stream(dev)
    start_ev = torch.cuda.Event()
    start_ev.record(cur)
    sf.wait_event(start_ev)
    st.wait_event(start_ev)
    # balanced RESNAP-aligned shard boundaries (must match the kernel's
    # own formula: floor(s*T/S) to the RESNAP grid)
    def shard_lo(s):
        return (s * T // nshards) // RESNAP * RESNAP if s < nshards else T

    # even shard groups: trades chunks cover equal time slices and every
    # flags launch keeps full occupancy. Both a small-first-group and a
    # geometric schedule were measured WORSE (209 / 199 vs 234 G/s): a
    # 1-shard flags launch runs at poor occupancy and delays the whole
    # pipeline, and geometric sizing parks half the timeline behind the
    # final flags launch.
    sbounds = [nshards * g // time_groups for g in range(time_groups + 1)]
    evs = []
    for g in range(time_groups):
        s0, s1 = sbounds[g], sbounds[g + 1]
        ops.bt_flags(cptr, pptr, eflags.data_ptr(), xflags.data_ptr(),
                     nsym, T, P, nshards, tail, s0, s1 - s0,
                     sf.cuda_stream)
        ev = torch.cuda.Event()
        ev.record(sf)
        evs.append(ev)
    for g in range(time_groups):
        s0, s1 = sbounds[g], sbounds[g + 1]
        t_lo = shard_lo(s0)
        t_hi = T if s1 == nshards else shard_lo(s1)
        st.wait_event(evs[g])
        ops.bt_trades(cptr, pptr, eflags.data_ptr(), xflags.data_ptr(),
                      metrics.data_ptr(), nsym, T, P,
                      float(initial_equity), 0, nsym, t_lo, t_hi,
                      carry.data_ptr(), st.cuda_stream)
    done = torch.cuda.Event()
    done.record(st)
    cur.wait_event(done)
    return metrics


def fitness_from_metrics(metrics: torch.Tensor) -> torch.Tensor:
    """Aggregate per-(param, symbol) fitness to per-param GA fitness:
    mean across symbols (the reference's GA evaluates one fitness per
    individual, strategy_evolution_service.py:525-694)."""
    return metrics[..., 9].mean(dim=1)
