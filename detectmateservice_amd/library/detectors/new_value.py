"""NewValueDetector family: set-membership anomaly detection.

Capability parity with the reference library's ``NewValueDetector``
(``detectors.new_value_detector``; config shapes
/root/reference/container/config/detector_config.yaml:1-9 and
tests/config/detector_config.yaml:1-17; behavior demo
docs/getting_started.md:421-511): watched fields are declared per
``global`` instance (applies to every event) or per ``events[event_id]``
instance; during the first ``data_use_training`` lines the detector learns
the values of watched fields; afterwards an unseen value raises an alert
with description ``"Unknown value: '<v>'"`` keyed like ``"Global - URL"``.

Watched-field specs: ``header_variables: [{pos: URL}]`` index
``ParserSchema.logFormatVariables`` by key; ``variables: [{pos: 0,
name: var1}]`` index ``ParserSchema.variables`` by position.

``NewValueComboDetector`` (reference docs/library.md:13) watches the
*tuple* of all configured fields instead of each field independently.

GPU path: the known-value sets are mirrored into a GPU hash set
(``detectmateservice_amd.ops.hashset``) and batches of parsed frames are
probed with one kernel launch; this Python implementation defines the
semantics the kernel is tested against.
"""
from __future__ import annotations

import time
from typing import Any, Dict, List, Optional, Set, Tuple

from pydantic import ConfigDict, Field

from ...components.base import CoreDetector, CoreDetectorConfig
from ...schemas import DetectorSchema, ParserSchema


class NewValueDetectorConfig(CoreDetectorConfig):
    model_config = ConfigDict(extra="allow", populate_by_name=True)

    method_type: str = "new_value_detector"
    auto_config: bool = False
    params: Dict[str, Any] = {}
    #: instance_name -> {"variables": [...], "header_variables": [...]}
    global_: Dict[str, Any] = Field(default_factory=dict, alias="global")
    #: event_id -> instance_name -> {...}
    events: Dict[Any, Any] = {}


class _WatchSpec:
    """One watched field: where it lives and how it is labeled in alerts."""

    __slots__ = ("scope", "event_id", "kind", "pos", "name")

    def __init__(self, scope: str, event_id: Optional[int], kind: str, pos, name: Optional[str]):
        self.scope = scope          # "Global" or "Event <id>"
        self.event_id = event_id    # None for global
        self.kind = kind            # "header" or "variable"
        self.pos = pos              # str key (header) or int index (variable)
        self.name = name or str(pos)

    @property
    def key(self) -> str:
        return f"{self.scope} - {self.name}"

    def extract(self, parsed: ParserSchema) -> Optional[str]:
        if self.kind == "header":
            return (parsed.logFormatVariables or {}).get(str(self.pos))
        try:
            idx = int(self.pos)
        except (TypeError, ValueError):
            return None
        variables = parsed.variables or []
        if 0 <= idx < len(variables):
            return variables[idx]
        return None


def _parse_specs(config: NewValueDetectorConfig) -> List[_WatchSpec]:
    specs: List[_WatchSpec] = []

    def add_block(scope: str, event_id: Optional[int], block: Dict[str, Any]) -> None:
        for var in block.get("variables") or []:
            specs.append(_WatchSpec(scope, event_id, "variable", var.get("pos"), var.get("name")))
        for var in block.get("header_variables") or []:
            specs.append(_WatchSpec(scope, event_id, "header", var.get("pos"), var.get("name")))

    for _instance, block in (config.global_ or {}).items():
        if isinstance(block, dict):
            add_block("Global", None, block)
    for event_id, instances in (config.events or {}).items():
        try:
            eid = int(event_id)
        except (TypeError, ValueError):
            continue
        for _instance, block in (instances or {}).items():
            if isinstance(block, dict):
                add_block(f"Event {eid}", eid, block)
    return specs


class NewValueDetector(CoreDetector):
    CONFIG_CLASS = NewValueDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        self.specs = _parse_specs(self.config)
        #: spec.key -> set of known values
        self.known: Dict[str, Set[str]] = {s.key: set() for s in self.specs}
        #: spec.key -> set of FNV-1a hashes (mirrors `known`; the batched
        #: C++ fast path compares hashes without building Python objects)
        self.known_h: Dict[str, Set[int]] = {s.key: set() for s in self.specs}
        self.detector_id = f"new_value_detector-{id(self):x}"
        self._learn_after_alert = bool(
            (self.config.params or {}).get("learn_after_alert", False)
        )

    def _learn(self, key: str, value: str) -> None:
        from ... import ops as _ops

        self.known[key].add(value)
        h = _ops.fnv1a64(value.encode("utf-8"))
        if h >= 1 << 63:  # two's-complement view: the C++ path returns int64
            h -= 1 << 64
        self.known_h[key].add(h)

    # ------------------------------------------------------------------
    # batched service-path fast path: C++ decode+hash of N frames in one
    # call (ops/csrc/codec.cpp::parser_watch_hashes) — the per-message
    # Python deserialization measured ~200 us/line in service mode.
    # ------------------------------------------------------------------
    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]:
        from ... import ops as _ops

        from ...components.base import BufferMode

        mode = getattr(self.config, "buffer_mode", BufferMode.NO_BUF)
        if isinstance(mode, str):
            mode = BufferMode(mode)
        if (
            len(frames) < 8
            or not _ops.have_extension()
            or not self.specs
            or mode != BufferMode.NO_BUF  # windowed modes need per-frame flow
        ):
            return super().process_batch(frames)
        from ...ops import _dmx_C  # type: ignore[attr-defined]

        var_specs = [
            (s.event_id if s.event_id is not None else -1, int(s.pos))
            for s in self.specs if s.kind == "variable"
        ]
        hdr_names = [str(s.pos) for s in self.specs if s.kind == "header"]
        col_keys = (
            [s.key for s in self.specs if s.kind == "variable"]
            + [s.key for s in self.specs if s.kind == "header"]
        )
        # header specs scoped to an event cannot be filtered in C++ (the
        # extractor applies header watches to every frame); fall back when
        # such a spec exists.
        if any(s.kind == "header" and s.event_id is not None for s in self.specs):
            return super().process_batch(frames)

        hashes, _event_ids, _log_ids = _dmx_C.parser_watch_hashes(
            list(frames), var_specs, hdr_names, False
        )
        results: List[Optional[bytes]] = [None] * len(frames)

        n_train = int(getattr(self.config, "data_use_training", 0))
        train_upto = 0
        if self._seen_lines < n_train:
            train_upto = min(len(frames), n_train - self._seen_lines)
            # training still records VALUES (state_dict readability +
            # python-path parity), so decode the training slice normally
            self.train([ParserSchema.deserialize(f) for f in frames[:train_upto]])
        self._seen_lines += len(frames)

        h = hashes.tolist()
        for i in range(train_upto, len(frames)):
            row = h[i]
            hit = any(
                v != 0 and v not in self.known_h[col_keys[w]]
                for w, v in enumerate(row)
            )
            if not hit:
                continue
            # rare path: full decode + the reference alert semantics
            parsed = ParserSchema.deserialize(frames[i])
            alert = DetectorSchema()
            if self.detect(parsed, alert):
                results[i] = alert.serialize()
        return results

    def _relevant_specs(self, parsed: ParserSchema) -> List[_WatchSpec]:
        return [
            s for s in self.specs
            if s.event_id is None or s.event_id == parsed.EventID
        ]

    def train(self, parsed_batch: List[ParserSchema]) -> None:
        for parsed in parsed_batch:
            for spec in self._relevant_specs(parsed):
                value = spec.extract(parsed)
                if value is not None:
                    self._learn(spec.key, value)

    def detect(self, parsed: ParserSchema, alert: DetectorSchema) -> bool:
        anomalies: Dict[str, str] = {}
        for spec in self._relevant_specs(parsed):
            value = spec.extract(parsed)
            if value is None:
                continue
            if value not in self.known[spec.key]:
                anomalies[spec.key] = value
                if self._learn_after_alert:
                    self._learn(spec.key, value)
        if not anomalies:
            return False
        first_val = next(iter(anomalies.values()))
        alert.detectorID = self.detector_id
        alert.detectorType = "new_value_detector"
        alert.alertID = f"nv-{parsed.logID or parsed.parsedLogID}"
        alert.detectionTimestamp = int(time.time())
        alert.logIDs = [parsed.logID] if parsed.logID else []
        alert.score = 1.0
        alert.description = f"Unknown value: '{first_val}'"
        alert.alertsObtain = {k: f"Unknown value: '{v}'" for k, v in anomalies.items()}
        return True

    # -- data-parallel state merge (dist_mode "dp") --------------------
    def dist_sync(self, group=None) -> None:
        """COLLECTIVE union of every rank's learned value sets: N
        data-parallel detector ranks each see a shard of the stream;
        after sync each knows every rank's values (so a value learned on
        rank 0 does not alert on rank 1). Low-frequency — end of the
        training phase or an operator's /admin/dp-sync."""
        import torch.distributed as tdist

        if not tdist.is_initialized() or tdist.get_world_size(group) <= 1:
            return
        world = tdist.get_world_size(group)
        local = {k: sorted(v) for k, v in self.known.items()}
        gathered: list = [None] * world
        tdist.all_gather_object(gathered, local, group=group)
        me = tdist.get_rank(group)
        for r, other in enumerate(gathered):
            if r == me or not other:
                continue
            for key, values in other.items():
                if key not in self.known:
                    self.known[key] = set()
                    self.known_h[key] = set()
                for v in values:
                    self._learn(key, v)

    # -- checkpoint/resume ----------------------------
    def state_dict(self) -> Dict[str, Any]:
        return {
            "seen_lines": self._seen_lines,
            "known": {k: sorted(v) for k, v in self.known.items()},
        }

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self._seen_lines = int(state.get("seen_lines", 0))
        for k, values in (state.get("known") or {}).items():
            for v in values:
                self.known.setdefault(k, set())
                self.known_h.setdefault(k, set())
                self._learn(k, v)


class NewValueComboDetector(CoreDetector):
    """Flags unseen *combinations* of the watched fields (docs/library.md:13)."""

    CONFIG_CLASS = NewValueDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        self.specs = _parse_specs(self.config)
        self.known_combos: Set[Tuple[Optional[str], ...]] = set()
        self.detector_id = f"new_value_combo_detector-{id(self):x}"

    def _combo(self, parsed: ParserSchema) -> Tuple[Optional[str], ...]:
        return tuple(
            s.extract(parsed)
            for s in self.specs
            if s.event_id is None or s.event_id == parsed.EventID
        )

    def train(self, parsed_batch: List[ParserSchema]) -> None:
        for parsed in parsed_batch:
            self.known_combos.add(self._combo(parsed))

    def detect(self, parsed: ParserSchema, alert: DetectorSchema) -> bool:
        combo = self._combo(parsed)
        if not combo or combo in self.known_combos:
            return False
        alert.detectorID = self.detector_id
        alert.detectorType = "new_value_combo_detector"
        alert.alertID = f"nvc-{parsed.logID or parsed.parsedLogID}"
        alert.detectionTimestamp = int(time.time())
        alert.logIDs = [parsed.logID] if parsed.logID else []
        alert.score = 1.0
        alert.description = f"Unknown combination: {combo!r}"
        return True

    def state_dict(self) -> Dict[str, Any]:
        return {
            "seen_lines": self._seen_lines,
            "known_combos": sorted("\x1f".join(x or "\x00" for x in c) for c in self.known_combos),
        }

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self._seen_lines = int(state.get("seen_lines", 0))
        for packed in state.get("known_combos") or []:
            self.known_combos.add(
                tuple(None if x == "\x00" else x for x in packed.split("\x1f"))
            )
