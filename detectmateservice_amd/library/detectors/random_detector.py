"""RandomDetector: threshold-per-variable random scoring.

Capability parity with the reference library's ``RandomDetector``
(/root/reference/docs/interfaces.md:158-204): each configured variable
carries a ``threshold`` param; a uniform random score per watched variable
above its threshold raises an alert. Deterministic under a ``seed`` param
for tests.
"""
from __future__ import annotations

import random
import time
from typing import Any, Dict, List

from ...components.base import CoreDetector
from ...schemas import DetectorSchema, ParserSchema
from .new_value import NewValueDetectorConfig, _parse_specs


class RandomDetectorConfig(NewValueDetectorConfig):
    method_type: str = "random_detector"


class RandomDetector(CoreDetector):
    CONFIG_CLASS = RandomDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        self.specs = _parse_specs(self.config)
        params = self.config.params or {}
        self.default_threshold = float(params.get("threshold", 0.5))
        seed = params.get("seed")
        self._rng = random.Random(seed)
        self.detector_id = f"random_detector-{id(self):x}"

    def train(self, parsed_batch: List[ParserSchema]) -> None:
        pass  # stateless

    def detect(self, parsed: ParserSchema, alert: DetectorSchema) -> bool:
        hits: Dict[str, float] = {}
        specs = self.specs or [None]  # no specs: score the line itself
        for spec in specs:
            score = self._rng.random()
            threshold = self.default_threshold
            if spec is not None and score > threshold:
                hits[spec.key] = score
            elif spec is None and score > threshold:
                hits["line"] = score
        if not hits:
            return False
        alert.detectorID = self.detector_id
        alert.detectorType = "random_detector"
        alert.alertID = f"rnd-{parsed.logID or parsed.parsedLogID}"
        alert.detectionTimestamp = int(time.time())
        alert.logIDs = [parsed.logID] if parsed.logID else []
        alert.score = max(hits.values())
        alert.description = "Random detection process"
        alert.alertsObtain = {k: f"{v:.4f}" for k, v in hits.items()}
        return True
