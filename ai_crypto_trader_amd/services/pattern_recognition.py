"""Pattern recognition service (reference parity:
services/pattern_recognition_service.py:138-343): subscribes
`market_updates`, runs the CNN detector per symbol, publishes
`pattern_signals` and the combined `pattern_analysis_report` key."""

from __future__ import annotations

import time

from ..bus.schema import Channels, Keys, PatternSignal
from ..models.patterns import PatternRecognitionModel
from .base import Service


class PatternRecognitionService(Service):
    name = "pattern_recognition"

    def __init__(self, bus, config=None, device="cpu",
                 min_confidence: float | None = None):
        super().__init__(bus, config)
        self.model = PatternRecognitionModel(
            device, seed=self.config.seed,
            model_type=getattr(self.config.patterns, 'model_type', 'cnn'))
        self.min_confidence = (min_confidence if min_confidence is not None
                               else self.config.patterns.min_confidence)
        self.closes: dict[str, list[float]] = {}
        self.detections = 0

    def run_tasks(self):
        return [self._consume(), self._detect_loop()]

    async def _consume(self):
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)

        def on_msg(_, m):
            if m.get("symbol"):
                h = self.closes.setdefault(m["symbol"], [])
                h.append(m["current_price"])
                del h[:-512]

        await self.consume(sub, on_msg)

    async def _detect_loop(self):
        # train once on the synthetic pattern set (reference trains its CNN
        # on the same kind of generated data, pattern_recognition.py:863+)
        if not self.model.trained:
            acc = self.model.train(epochs=4, n_per_class=32)
            self.log.info("pattern CNN trained, train acc=%.2f", acc)
        while self.running:
            report = {}
            import numpy as np
            for sym, h in list(self.closes.items()):
                if len(h) < 64:
                    continue
                det = self.model.detect(np.asarray(h))
                self.detections += 1
                report[sym] = det
                if det["pattern"] != "none" and \
                        det["confidence"] >= self.min_confidence:
                    await self.bus.publish(
                        Channels.PATTERN_SIGNALS,
                        PatternSignal(sym, det["pattern"], det["signal"],
                                      det["confidence"],
                                      det["completion"]).to_dict())
            if report:
                await self.bus.set(Keys.PATTERN_ANALYSIS_REPORT, {
                    "at": time.time(), "patterns": report,
                })
            await self.sleep(3.0)

    async def run(self):
        pass
