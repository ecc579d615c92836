"""OutputAggregator: DetectorSchema alerts → OutputSchema egress frames.

Capability parity with the reference's fluentout stage
(/root/reference/container/fluentout/fluent.conf + schemas_pb.rb —
SURVEY.md §2.3 OutputSchema, §3.5 demo data path): collects detector
alerts and emits aggregated OutputSchema frames, either one per alert
(``window_size=1``, streaming passthrough like the fluentd file sink) or
batched windows. Optionally appends JSON lines to ``output_file`` exactly
like the shipped fluentd file output writes
``container/fluentlogs/output.%Y%m%d``.
"""
from __future__ import annotations

import json
import time
from pathlib import Path
from typing import Any, Dict, List, Optional

from ...components.base import CoreComponent, CoreConfig
from ...schemas import DetectorSchema, OutputSchema


class OutputAggregatorConfig(CoreConfig):
    method_type: str = "output_aggregator"
    window_size: int = 1
    output_file: Optional[str] = None


class OutputAggregator(CoreComponent):
    CONFIG_CLASS = OutputAggregatorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        self._window: List[DetectorSchema] = []

    def _emit(self) -> bytes:
        alerts = self._window
        self._window = []
        out = OutputSchema(
            detectorIDs=[a.detectorID for a in alerts],
            detectorTypes=[a.detectorType for a in alerts],
            alertIDs=[a.alertID for a in alerts],
            outputTimestamp=int(time.time()),
            logIDs=[lid for a in alerts for lid in (a.logIDs or [])],
            extractedTimestamps=[t for a in alerts for t in (a.extractedTimestamps or [])],
            description="; ".join(a.description for a in alerts if a.description),
            alertsObtain={k: v for a in alerts for k, v in (a.alertsObtain or {}).items()},
        )
        if self.config.output_file:
            path = Path(self.config.output_file)
            path.parent.mkdir(parents=True, exist_ok=True)
            with open(path, "a", encoding="utf-8") as fh:
                fh.write(json.dumps(out.to_dict(), default=str) + "\n")
        return out.serialize()

    def process(self, data: bytes) -> Optional[bytes]:
        alert = DetectorSchema.deserialize(data)
        self._window.append(alert)
        if len(self._window) >= max(1, int(self.config.window_size)):
            return self._emit()
        return None

    def state_dict(self) -> Dict[str, Any]:
        return {"window": [a.to_dict() for a in self._window]}

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self._window = [DetectorSchema(d) for d in state.get("window", [])]
