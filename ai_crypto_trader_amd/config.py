"""Schema-validated configuration.

The reference reads a 900-line config.json raw in every service
(`json.load` per __init__, e.g. neural_network_service.py:50-51) with no
schema or validation. Here the same configuration blocks
are pydantic models with defaults mirroring the reference's config.json
values, validated once and shared.
"""

from __future__ import annotations

import json
from pathlib import Path

from pydantic import BaseModel, Field


class TradingConfig(BaseModel):
    symbols: list[str] = Field(default_factory=lambda: ["BTCUSDC", "ETHUSDC"])
    quote_asset: str = "USDC"
    # venue adapter: "fake" (offline sim) or "binance" (live REST via
    # ai_crypto_trader_amd.live) — the only switch callers need
    exchange: str = "fake"
    ai_analysis_interval: float = 60.0        # config.json:10
    min_price_change_pct: float = 0.5
    min_volume_usdc: float = 10_000.0
    fee_rate: float = 0.001
    max_positions: int = 5
    min_confidence: float = 0.7
    min_trade_usd: float = 40.0               # trading_strategy.md minimum


class RiskConfig(BaseModel):
    var_confidence: float = 0.95
    lookback_days: int = 30
    max_portfolio_var_pct: float = 0.10       # alert_rules.yml threshold
    position_sizing: str = "equal_risk"       # equal_risk | half_kelly | fixed
    fixed_position_pct: float = 0.1
    adaptive_stop_vol_factor_min: float = 0.5
    adaptive_stop_vol_factor_max: float = 2.0
    base_stop_loss_pct: float = 0.02
    correlation_threshold: float = 0.7
    max_daily_drawdown_pct: float = 0.06      # trading_strategy.md halt


class MonteCarloConfig(BaseModel):
    simulation_method: str = "gbm"            # gbm | historical (:275-298)
    num_simulations: int = 10_000             # reference default 1000
    time_horizon_days: int = 30               # config.json:87-89
    n_assets: int = 16
    scenarios: dict[str, tuple[float, float]] = Field(
        # name -> (mu multiplier, sigma multiplier) (config.json:97-103)
        default_factory=lambda: {
            "base": (1.0, 1.0), "bull": (1.5, 0.8), "bear": (-1.0, 1.3),
            "volatile": (1.0, 2.0), "crab": (0.2, 0.6),
        }
    )
    interval_s: float = 3600.0


class NNConfig(BaseModel):
    seq_len: int = 60                         # config.json:410
    batch_size: int = 32
    epochs: int = 100
    n_features: int = 9
    hidden: tuple[int, int] = (64, 32)        # config.json:454-458
    lr: float = 1e-3
    retrain_interval_s: float = 86_400.0      # model_checkpoint_interval
    model_dir: str = "models_store"
    # NN feature attribution (SHAP stand-in): grad_input |
    # integrated_gradients (neural_network_service.py:957-1003)
    attribution: str = "integrated_gradients"


class EvolutionConfig(BaseModel):
    population_size: int = 1024               # reference: 20 (GA_POPULATION_SIZE)
    generations: int = 10
    elite_k: int = 16
    tournament: int = 4
    cx_rate: float = 0.5
    mut_rate: float = 0.15
    mut_scale: float = 0.1
    method: str = "hybrid"                    # ga | rl | gpt | hybrid
    min_sharpe_ratio: float = 1.2             # config.json:208
    max_drawdown: float = 0.15                # config.json:209
    min_win_rate: float = 0.52                # config.json:210
    min_profit_factor: float = 1.2            # config.json:211
    interval_s: float = 3600.0


class RegimeConfig(BaseModel):
    method: str = "kmeans"                    # kmeans | gmm | hmm | rf | rule
    n_regimes: int = 4
    lookback: int = 500
    retrain_interval_s: float = 3600.0


class GridConfig(BaseModel):
    levels: int = 10
    spacing: str = "arithmetic"               # arithmetic | geometric
    range_pct: float = 0.05
    order_size_pct: float = 0.1


class DCAConfig(BaseModel):
    schedule: str = "fixed"        # fixed | regime_based | value_averaging
    interval_s: float = 86_400.0
    base_order_usd: float = 100.0
    dip_threshold_pct: float = 0.05
    dip_multiplier: float = 2.0


class SocialConfig(BaseModel):
    # metrics source: synthetic (offline) | lunarcrush (live, needs
    # LUNARCRUSH_API_KEY)
    source: str = "synthetic"
    update_interval_s: float = 300.0
    sentiment_half_life_h: float = 6.0        # social_risk_adjuster half-life
    source_weights: dict[str, float] = Field(
        default_factory=lambda: {
            "twitter": 0.35, "reddit": 0.25, "news": 0.4,
        }
    )
    lead_lag_max_h: int = 24                  # social_metrics_analyzer +-24h


class BusConfig(BaseModel):
    backend: str = "inprocess"                # inprocess | redis
    redis_host: str = "localhost"
    redis_port: int = 6379


class BenchConfig(BaseModel):
    pop_per_gpu: int = 1024
    symbols: int = 64
    candles: int = 1_000_000
    mc_paths: int = 10_000_000
    mc_assets: int = 64


class ArbitrageConfig(BaseModel):
    min_profit_pct: float = 0.05              # reference config.json arbitrage_detection
    max_path_length: int = 3
    interval_s: float = 30.0


class OrderBookConfig(BaseModel):
    interval_s: float = 2.0                   # reference: 60 s against live Binance
    depth_levels: int = 20
    impact_trade_sizes: list[float] = [10_000, 50_000, 100_000, 500_000,
                                       1_000_000]


class NewsConfig(BaseModel):
    # headline source: synthetic (offline) | live (CryptoPanic + RSS +
    # LunarCrush; keys optional)
    source: str = "synthetic"
    interval_s: float = 2.0                   # reference: 300 s against live feeds
    max_items_per_source: int = 20


class PatternConfig(BaseModel):
    min_confidence: float = 0.5               # pattern_detection_threshold
    interval_s: float = 5.0
    # classifier architecture (reference model types,
    # pattern_recognition.py:94-196): cnn | lstm | cnn_lstm
    model_type: str = "cnn"


class VolumeProfileConfig(BaseModel):
    n_bins: int = 24
    value_area_pct: float = 0.70


class FeatureImportanceConfig(BaseModel):
    n_permutations: int = 30                  # reference config.json:315
    interval_s: float = 30.0


class AppConfig(BaseModel):
    trading: TradingConfig = Field(default_factory=TradingConfig)
    risk: RiskConfig = Field(default_factory=RiskConfig)
    monte_carlo: MonteCarloConfig = Field(default_factory=MonteCarloConfig)
    neural_network: NNConfig = Field(default_factory=NNConfig)
    evolution: EvolutionConfig = Field(default_factory=EvolutionConfig)
    regime: RegimeConfig = Field(default_factory=RegimeConfig)
    grid: GridConfig = Field(default_factory=GridConfig)
    dca: DCAConfig = Field(default_factory=DCAConfig)
    social: SocialConfig = Field(default_factory=SocialConfig)
    bus: BusConfig = Field(default_factory=BusConfig)
    bench: BenchConfig = Field(default_factory=BenchConfig)
    arbitrage: ArbitrageConfig = Field(default_factory=ArbitrageConfig)
    order_book: OrderBookConfig = Field(default_factory=OrderBookConfig)
    news: NewsConfig = Field(default_factory=NewsConfig)
    patterns: PatternConfig = Field(default_factory=PatternConfig)
    volume_profile: VolumeProfileConfig = Field(
        default_factory=VolumeProfileConfig)
    feature_importance: FeatureImportanceConfig = Field(
        default_factory=FeatureImportanceConfig)
    data_dir: str = "backtesting_data"
    log_dir: str = "logs"
    seed: int = 0

    @classmethod
    def load(cls, path: str | Path | None = None) -> "AppConfig":
        if path is None:
            for cand in (Path("configs/config.json"), Path("config.json")):
                if cand.exists():
                    path = cand
                    break
        if path is None:
            cfg = cls()
        else:
            with open(path) as f:
                cfg = cls.model_validate(json.load(f))
        return apply_env_overrides(cfg)

    def save(self, path: str | Path):
        Path(path).parent.mkdir(parents=True, exist_ok=True)
        with open(path, "w") as f:
            f.write(self.model_dump_json(indent=2))


# Env-var flag layer (reference parity: per-service env flags, e.g.
# EVOLUTION_METHOD / GA_POPULATION_SIZE strategy_evolution_service.py:75-79,
# selection weights strategy_selection_service.py:71-78, plus .env secrets).
# Two forms:
#   ACT_<section>__<field>=value  — generic override for any config field
#   legacy aliases below          — the reference's documented flag names
_ENV_ALIASES = {
    "EVOLUTION_METHOD": ("evolution", "method"),
    "GA_POPULATION_SIZE": ("evolution", "population_size"),
    "GA_GENERATIONS": ("evolution", "generations"),
    "MIN_CONFIDENCE": ("trading", "min_confidence"),
    "TRADING_SYMBOLS": ("trading", "symbols"),
    "MC_NUM_SIMULATIONS": ("monte_carlo", "num_simulations"),
    "REGIME_METHOD": ("regime", "method"),
    "SEED": (None, "seed"),
}


def _coerce(cur, raw: str):
    if isinstance(cur, bool):
        return raw.lower() in ("1", "true", "yes", "on")
    if isinstance(cur, int):
        return int(raw)
    if isinstance(cur, float):
        return float(raw)
    if isinstance(cur, list):
        return [s.strip() for s in raw.split(",") if s.strip()]
    return raw


def apply_env_overrides(cfg: "AppConfig") -> "AppConfig":
    import os

    def set_field(section, field, raw):
        target = cfg if section is None else getattr(cfg, section, None)
        if target is None or not hasattr(target, field):
            return
        setattr(target, field, _coerce(getattr(target, field), raw))

    for name, (section, field) in _ENV_ALIASES.items():
        if name in os.environ:
            set_field(section, field, os.environ[name])
    for name, raw in os.environ.items():
        if name.startswith("ACT_") and "__" in name:
            section, _, field = name[4:].lower().partition("__")
            set_field(section, field, raw)
    return cfg


_global: AppConfig | None = None


def get_config() -> AppConfig:
    global _global
    if _global is None:
        _global = AppConfig.load()
    return _global


def set_config(cfg: AppConfig):
    global _global
    _global = cfg
