"""Typed payload schema for the reference's channels and keys.

The reference's ~20 pub/sub channels and ~70 keys carry loosely-specified
JSON dicts (SURVEY.md §1.1 catalog). These dataclasses schema-lock the
payload shapes: channel names and field names reproduce the reference
verbatim (citations per class), so a consumer written against the
reference's schema can read this bus.
"""

from __future__ import annotations

import time
from dataclasses import asdict, dataclass, field


def now_iso() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime())


class Channels:
    """Channel-name constants (producer -> consumers map in SURVEY.md §1.1)."""

    MARKET_UPDATES = "market_updates"                 # market_monitor_service.py:533
    TRADING_OPPORTUNITIES = "trading_opportunities"   # :563-574
    SOCIAL_UPDATES = "social_updates"                 # social_monitor_service.py:234
    TRADING_SIGNALS = "trading_signals"               # ai_analyzer_service.py:627
    RISK_ENRICHED_SIGNALS = "risk_enriched_signals"   # portfolio_risk_service.py:847
    STRATEGY_UPDATE = "strategy_update"               # strategy_evolution_service.py:356
    STRATEGY_SWITCH = "strategy_switch"               # market_regime_service.py:1034
    STRATEGY_EVOLUTION_UPDATES = "strategy_evolution_updates"  # :1703
    STOP_LOSS_ADJUSTMENTS = "stop_loss_adjustments"   # portfolio_risk_service.py:707
    RISK_ALERTS = "risk_alerts"                       # :649, :791
    NN_PREDICTIONS = "neural_network_predictions"     # neural_network_service.py:1394
    NN_EVENTS = "neural_network_events"               # :1499
    PATTERN_SIGNALS = "pattern_signals"               # pattern_recognition_service.py:216
    FEATURE_IMPORTANCE = "feature_importance"         # feature_importance_analyzer.py:610
    MODEL_REGISTRY_EVENTS = "model_registry_events"   # model_registry_service.py:168
    MODEL_PERFORMANCE_UPDATES = "model_performance_updates"    # :221-273
    NEWS_ANALYSIS_UPDATES = "news_analysis_updates"
    ENHANCED_SOCIAL_UPDATES = "enhanced_social_updates"
    RISK_ADJUSTMENT_UPDATES = "risk_adjustment_updates"
    EXPLAINED_TRADING_SIGNALS = "explained_trading_signals"
    GRID_TRADE_NOTIFICATIONS = "grid_trade_notifications"
    DCA_PURCHASE_NOTIFICATIONS = "dca_purchase_notifications"
    ARBITRAGE_NOTIFICATIONS = "arbitrage_notifications"
    CONFIG_UPDATES = "config_updates"
    # dashboard-side channels (subscribed at dashboard.py:91-99)
    TRADE_EXECUTIONS = "trade_executions"
    PORTFOLIO_UPDATES = "portfolio_updates"
    AI_MODEL_UPDATES = "ai_model_updates"
    RISK_METRICS_UPDATES = "risk_metrics_updates"


class Keys:
    """KV / hash key names (SURVEY.md §1.1 key catalog)."""

    HOLDINGS = "holdings"                              # trade_executor_service.py:709
    ACTIVE_TRADES = "active_trades"                    # :1212
    TRAILING_STOPS = "trailing_stops"                  # :1126
    CURRENT_PRICES = "current_prices"                  # market_monitor_service.py:541
    SOCIAL_METRICS = "social_metrics"                  # social_monitor_service.py:244
    PORTFOLIO_RISK = "portfolio_risk"                  # portfolio_risk_service.py:630
    ADAPTIVE_STOP_LOSSES = "adaptive_stop_losses"      # :713
    PORTFOLIO_DIVERSIFICATION = "portfolio_diversification"    # :774
    MONTE_CARLO_RESULTS = "monte_carlo_results"        # monte_carlo_service.py:564
    MONTE_CARLO_LATEST_REPORT = "monte_carlo_latest_report"    # :905
    MONTE_CARLO_REQUEST = "monte_carlo_request"        # :783
    MARKET_REGIME_HISTORY = "market_regime_history"
    CURRENT_MARKET_REGIME = "current_market_regime"    # strategy_evolution:262
    STRATEGY_PARAMS = "strategy_params"                # :353
    SOCIAL_RISK_ADJUSTMENTS = "social_risk_adjustments"
    SOCIAL_RISK_REPORT = "social_risk_report"
    PATTERN_ANALYSIS_REPORT = "pattern_analysis_report"
    NEWS_ANALYSIS = "news_analysis"                    # ai_analyzer_service.py:429
    NEWS_SUMMARY_REPORT = "news_summary_report"
    FEATURE_IMPORTANCE = "feature_importance"
    MODEL_REGISTRY = "model_registry"

    @staticmethod
    def nn_prediction(symbol: str, interval: str) -> str:
        return f"nn_prediction_{symbol}_{interval}"    # nn_service.py:1202

    @staticmethod
    def historical_data(symbol: str, interval: str) -> str:
        return f"historical_data_{symbol}_{interval}"  # nn_service.py:501

    @staticmethod
    def order_book(symbol: str) -> str:
        return f"order_book:{symbol}"

    @staticmethod
    def order_book_agg(symbol: str) -> str:
        return f"order_book_agg:{symbol}"

    @staticmethod
    def grid_config(symbol: str) -> str:
        return f"grid_config:{symbol}"

    @staticmethod
    def strategy_performance(strategy_id: str) -> str:
        return f"strategy_performance_{strategy_id}"

    @staticmethod
    def strategy_trades(strategy_id: str) -> str:
        return f"strategy_trades_{strategy_id}"

    @staticmethod
    def social_lead_lag(symbol: str) -> str:
        return f"social_lead_lag:{symbol}"              # social_metrics:442

    GRID_PERFORMANCE = "grid_performance"
    DCA_PERFORMANCE = "dca_performance"
    DCA_PURCHASE_LIST = "dca_purchase_list"
    ARBITRAGE_OPPORTUNITIES = "arbitrage_opportunities"
    SELECTED_STRATEGY = "selected_strategy"
    MC_FAN_CHART = "monte_carlo_fan_chart"


@dataclass
class MarketUpdate:
    """`market_updates` payload (market_monitor_service.py:445-524; field
    list documented in the reference README.md:352-374)."""

    symbol: str
    current_price: float
    avg_volume: float
    timestamp: str = field(default_factory=now_iso)
    rsi: float = 50.0
    rsi_3m: float = 50.0
    rsi_5m: float = 50.0
    stoch_k: float = 50.0
    macd: float = 0.0
    macd_3m: float = 0.0
    macd_5m: float = 0.0
    williams_r: float = -50.0
    bb_position: float = 0.5
    trend: str = "neutral"
    trend_strength: float = 0.0
    price_change_1m: float = 0.0
    price_change_3m: float = 0.0
    price_change_5m: float = 0.0
    price_change_15m: float = 0.0
    combined_indicators: dict = field(default_factory=dict)
    volume_profile: dict = field(default_factory=dict)

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class SocialMetricsBlock:
    # field names verbatim per reference README.md:377-401
    social_volume: float = 0.0
    social_engagement: float = 0.0
    social_contributors: float = 0.0
    social_sentiment: float = 0.5
    twitter_volume: float = 0.0
    reddit_volume: float = 0.0
    news_volume: float = 0.0


@dataclass
class SocialUpdate:
    """`social_updates` payload (social_monitor_service.py:234-241;
    README.md:377-401)."""

    symbol: str
    metrics: SocialMetricsBlock = field(default_factory=SocialMetricsBlock)
    weighted_sentiment: float = 0.5
    recent_news: list = field(default_factory=list)
    timestamp: str = field(default_factory=now_iso)

    def to_dict(self) -> dict:
        return {
            "symbol": self.symbol,
            "data": {
                "metrics": asdict(self.metrics),
                "weighted_sentiment": self.weighted_sentiment,
                "recent_news": self.recent_news,
                "timestamp": self.timestamp,
            },
        }


@dataclass
class TradingSignal:
    """`trading_signals` payload (ai_trader.py:108-167; README.md:516-575)."""

    symbol: str
    decision: str                     # BUY | SELL | HOLD
    confidence: float
    reasoning: str = ""
    risk_level: str = "medium"
    key_indicators: list = field(default_factory=list)
    social_impact: str = ""                      # README.md:525
    selected_strategy: dict = field(default_factory=dict)   # :529-543
    explanation: dict = field(default_factory=dict)
    factor_weights: dict = field(default_factory=dict)
    model_version: str = "rule-1.0"
    model_id: str = "local_analyst"
    market_data: dict = field(default_factory=dict)
    timestamp: str = field(default_factory=now_iso)

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class RiskInfo:
    """risk_info block added by portfolio risk enrichment
    (portfolio_risk_service.py:808-846)."""

    var: float = 0.0
    var_pct: float = 0.0
    cvar: float = 0.0
    portfolio_var: float = 0.0
    optimal_position_pct: float = 0.1
    adaptive_stop_loss: float = 0.0
    adaptive_stop_pct: float = 0.02

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class StrategySwitch:
    """`strategy_switch` payload (market_regime_service.py:1034)."""

    market_regime: str
    old_strategy_id: str
    new_strategy_id: str
    reason: str
    timestamp: str = field(default_factory=now_iso)

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class StopLossAdjustment:
    """`stop_loss_adjustments` payload (portfolio_risk_service.py:707)."""

    symbol: str
    current_stop_price: float
    recommended_stop_price: float
    reason: str
    details: dict = field(default_factory=dict)

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class RiskAlert:
    """`risk_alerts` payload (portfolio_risk_service.py:649, :791)."""

    type: str            # portfolio_var_exceeded | poor_diversification
    detail: dict = field(default_factory=dict)
    timestamp: str = field(default_factory=now_iso)

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class NNPrediction:
    """`neural_network_predictions` payload (neural_network_service.py:1394;
    README.md:488-513)."""

    symbol: str
    interval: str
    predicted_price: float
    current_price: float
    predicted_change_pct: float
    confidence: float
    model_type: str = "lstm"
    status: str = "success"                       # README.md:499
    prediction_time: str = field(default_factory=now_iso)
    reference_time: str = field(default_factory=now_iso)
    training_metrics: dict = field(default_factory=dict)      # :501-505
    features_used: list = field(default_factory=list)         # :506-512
    timestamp: str = field(default_factory=now_iso)

    def to_dict(self) -> dict:
        d = asdict(self)
        d["change_pct"] = self.predicted_change_pct   # reference name
        return d


@dataclass
class PatternSignal:
    """`pattern_signals` payload (pattern_recognition_service.py:216)."""

    symbol: str
    pattern: str
    signal: str            # bullish | bearish | neutral
    strength: float
    completion: float
    timestamp: str = field(default_factory=now_iso)

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class EvolutionUpdate:
    """`strategy_evolution_updates` (strategy_evolution_service.py:1703)."""

    strategy_id: str
    new_params: dict
    performance: dict
    risk_level: str = "medium"
    market_regime: str = "unknown"
    timestamp: str = field(default_factory=now_iso)

    def to_dict(self) -> dict:
        return asdict(self)
