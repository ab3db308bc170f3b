#!/usr/bin/env python3
"""Backtesting CLI (reference parity: run_backtest.py:24-227 — subcommands
fetch / backtest / list / analyze, plus `optimize` for GA parameter search
over the GPU backtest kernel).

Examples:
  python run_backtest.py fetch --symbol BTCUSDC --candles 20000
  python run_backtest.py backtest --symbol BTCUSDC --strategy dca_strategy
  python run_backtest.py backtest --symbol BTCUSDC --strategy momentum --plot
  python run_backtest.py optimize --symbol BTCUSDC --pop 256 --generations 10
  python run_backtest.py list
  python run_backtest.py analyze [--metric sharpe_ratio]
"""

from __future__ import annotations

import argparse
import json

from ai_crypto_trader_amd.backtesting.data_manager import (
    HistoricalDataManager,
)
from ai_crypto_trader_amd.backtesting.engine import (
    STRATEGY_PRESETS, BacktestEngine,
)
from ai_crypto_trader_amd.backtesting.result_analyzer import ResultAnalyzer


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--data-dir", default="backtesting_data")
    sub = ap.add_subparsers(dest="cmd", required=True)

    f = sub.add_parser("fetch", help="fetch (synthesize) historical data")
    f.add_argument("--symbol", required=True)
    f.add_argument("--interval", default="1m")
    f.add_argument("--candles", type=int, default=10_000)
    f.add_argument("--social", action="store_true")

    b = sub.add_parser("backtest", help="run a backtest")
    b.add_argument("--symbol", required=True)
    b.add_argument("--strategy", default="default",
                   choices=sorted(STRATEGY_PRESETS))
    b.add_argument("--interval", default="1m")
    b.add_argument("--candles", type=int, default=10_000)
    b.add_argument("--cpu", action="store_true",
                   help="force the CPU reference engine")
    b.add_argument("--plot", action="store_true")
    b.add_argument("--params", type=str, default=None,
                   help="JSON dict of parameter overrides")

    o = sub.add_parser("optimize", help="GA parameter search")
    o.add_argument("--symbol", required=True)
    o.add_argument("--pop", type=int, default=256)
    o.add_argument("--generations", type=int, default=10)
    o.add_argument("--candles", type=int, default=20_000)
    o.add_argument("--cpu", action="store_true")

    sub.add_parser("list", help="list stored data and results")
    an = sub.add_parser("analyze",
                        help="summary report over stored results")
    an.add_argument("--metric", default=None,
                    help="rank results by this metric only (reference "
                         "run_backtest.py `analyze --metric "
                         "sharpe_ratio`); aliases: sharpe_ratio, "
                         "total_return, win_rate, profit_factor")

    args = ap.parse_args()
    dm = HistoricalDataManager(args.data_dir)

    if args.cmd == "fetch":
        df = dm.fetch_market_data(args.symbol, args.interval, args.candles)
        print(f"fetched {len(df)} {args.interval} candles for "
              f"{args.symbol} -> {dm.market_path(args.symbol, args.interval)}")
        if args.social:
            sdf = dm.fetch_social_data(args.symbol)
            print(f"fetched {len(sdf)} social rows -> "
                  f"{dm.social_path(args.symbol)}")
        return

    if args.cmd == "backtest":
        eng = BacktestEngine(args.data_dir,
                             device="cpu" if args.cpu else None)
        params = json.loads(args.params) if args.params else None
        stats = eng.run_backtest(
            args.symbol, args.strategy, args.interval, args.candles,
            params=params, record_equity=args.plot)
        if args.plot:
            p = ResultAnalyzer(f"{args.data_dir}/analysis").plot_equity_curve(
                stats)
            print(f"plot -> {p}")
            stats.pop("equity_curve", None)
        print(json.dumps(stats, indent=2, default=str))
        return

    if args.cmd == "optimize":
        eng = BacktestEngine(args.data_dir,
                             device="cpu" if args.cpu else None)
        stats = eng.optimize(args.symbol, args.pop, args.generations,
                             args.candles)
        print(json.dumps(stats, indent=2, default=str))
        return

    if args.cmd == "list":
        eng = BacktestEngine(args.data_dir, device="cpu")
        avail = dm.list_available()
        print(json.dumps({
            "data": avail,
            "results": [
                {k: r.get(k) for k in ("symbol", "strategy", "sharpe",
                                       "total_return_pct", "n_trades")}
                for r in eng.list_results()
            ],
        }, indent=2))
        return

    if args.cmd == "analyze":
        _METRIC_ALIASES = {"sharpe_ratio": "sharpe",
                           "total_return": "total_return_pct",
                           "return": "total_return_pct"}
        eng = BacktestEngine(args.data_dir, device="cpu")
        results = eng.list_results()
        ra = ResultAnalyzer(f"{args.data_dir}/analysis")
        report = ra.summary_report(results)
        if args.metric:
            m = _METRIC_ALIASES.get(args.metric, args.metric)
            best = report.get("best_by", {}).get(m)
            if best is None:
                known = sorted(report.get("best_by", {}))
                raise SystemExit(f"unknown metric {args.metric!r} "
                                 f"(known: {known})")
            print(json.dumps({"metric": m, "best": best,
                              "n_results": report.get("n_results")},
                             indent=2))
        else:
            print(json.dumps(report, indent=2))
        if results:
            p = ra.comparison_chart(results)
            print(f"comparison chart -> {p}")


if __name__ == "__main__":
    main()
