#!/usr/bin/env python3
"""Flag-driven launcher for the model-management services (reference
parity: run_ai_model_services.py:29-107 — `--model-registry`,
`--explainability`, plus `--feature-importance` and `--patterns`).

  python run_ai_model_services.py --model-registry --explainability
"""

from __future__ import annotations

import argparse
import asyncio

from ai_crypto_trader_amd.bus.message_bus import InProcessBus
from ai_crypto_trader_amd.config import AppConfig
from ai_crypto_trader_amd.services.pattern_recognition import (
    PatternRecognitionService,
)
from ai_crypto_trader_amd.services.registry import (
    AIExplainabilityService, FeatureImportanceAnalyzer, ModelRegistryService,
)


async def run_services(args) -> None:
    bus = InProcessBus()
    cfg = AppConfig()
    services = []
    if args.model_registry:
        services.append(ModelRegistryService(bus, cfg))
    if args.explainability:
        services.append(AIExplainabilityService(bus, cfg))
    if args.feature_importance:
        services.append(FeatureImportanceAnalyzer(bus, cfg))
    if args.patterns:
        services.append(PatternRecognitionService(bus, cfg))
    if not services:
        print("nothing selected; see --help")
        return
    for s in services:
        await s.start()
    print(f"running: {', '.join(s.name for s in services)}")
    try:
        await asyncio.sleep(args.seconds)
    finally:
        for s in services:
            await s.stop()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model-registry", action="store_true")
    ap.add_argument("--explainability", action="store_true")
    ap.add_argument("--feature-importance", action="store_true")
    ap.add_argument("--patterns", action="store_true")
    ap.add_argument("--seconds", type=float, default=3600.0)
    asyncio.run(run_services(ap.parse_args()))


if __name__ == "__main__":
    main()
