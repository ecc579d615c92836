"""DummyDetector test double.

Behavior parity with the reference library's
``detectmatelibrary_tests.test_detectors.dummy_detector.DummyDetector``
(observed via /root/reference/tests/library_integration/
test_detector_integration.py:89-115): alternates False, True, False, ...;
score 1.0; description "Dummy detection process".
"""
from __future__ import annotations

import time
from typing import List

from ...components.base import CoreDetector, CoreDetectorConfig
from ...schemas import DetectorSchema, ParserSchema


class DummyDetectorConfig(CoreDetectorConfig):
    method_type: str = "dummy_detector"


class DummyDetector(CoreDetector):
    CONFIG_CLASS = DummyDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        self._calls = 0

    def train(self, parsed_batch: List[ParserSchema]) -> None:
        pass

    def detect(self, parsed: ParserSchema, alert: DetectorSchema) -> bool:
        fire = self._calls % 2 == 1  # False, True, False, True ...
        self._calls += 1
        if not fire:
            return False
        alert.detectorID = "dummy_detector"
        alert.detectorType = "dummy_detector"
        alert.alertID = f"dummy-{self._calls}"
        alert.detectionTimestamp = int(time.time())
        alert.logIDs = [parsed.logID] if parsed.logID else []
        alert.score = 1.0
        alert.description = "Dummy detection process"
        return True
