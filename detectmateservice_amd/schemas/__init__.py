"""Inter-stage wire schemas (proto3-compatible).

Message definitions match the reference's compiled descriptor
(/root/reference/container/fluentout/schemas_pb.rb:9; field table in
SURVEY.md §2.3) so frames interoperate with the reference's fluentd
plugins. Wrapper classes expose ``serialize()`` / ``deserialize(bytes)``
and dict-style construction, matching how the reference's tests build
frames (tests/library_integration/library_integration_base_fixtures.py:80-84).
"""
from __future__ import annotations

from typing import Any, ClassVar, Dict, Tuple

from . import codec
from .codec import (
    STRING,
    INT32,
    FLOAT,
    REP_STRING,
    REP_INT32,
    MAP_SS,
    default_for,
)

SCHEMA_VERSION = "0.3"


class BaseSchema:
    """Base for all wire messages: spec-driven encode/decode + dict access."""

    FIELDS: ClassVar[Dict[str, Tuple[int, str]]] = {"__version__": (1, STRING)}

    def __init__(self, data: Dict[str, Any] | None = None, **kwargs: Any) -> None:
        values = dict(data or {})
        values.update(kwargs)
        for name, (_num, kind) in self.FIELDS.items():
            setattr(self, name, values.pop(name, default_for(kind)))
        if values:
            raise ValueError(
                f"unknown fields for {type(self).__name__}: {sorted(values)}"
            )
        if not getattr(self, "__version__", ""):
            setattr(self, "__version__", SCHEMA_VERSION)

    # -- wire ----------------------------------------------------------
    def serialize(self) -> bytes:
        return codec.encode_message(
            self.FIELDS, {n: getattr(self, n) for n in self.FIELDS}
        )

    @classmethod
    def deserialize(cls, data: bytes) -> "BaseSchema":
        return cls(codec.decode_message(cls.FIELDS, data))

    # -- ergonomics ----------------------------------------------------
    def to_dict(self) -> Dict[str, Any]:
        return {n: getattr(self, n) for n in self.FIELDS}

    def __getitem__(self, key: str) -> Any:
        if key not in self.FIELDS:
            raise KeyError(key)
        return getattr(self, key)

    def __setitem__(self, key: str, value: Any) -> None:
        if key not in self.FIELDS:
            raise KeyError(key)
        setattr(self, key, value)

    def __eq__(self, other: object) -> bool:
        return isinstance(other, type(self)) and self.to_dict() == other.to_dict()

    def __repr__(self) -> str:  # pragma: no cover - debug aid
        nonzero = {
            n: getattr(self, n)
            for n in self.FIELDS
            if getattr(self, n) not in ("", 0, 0.0, [], {})
        }
        return f"{type(self).__name__}({nonzero})"


class Schema(BaseSchema):
    """``Schema { string __version__ = 1; }``"""


class LogSchema(BaseSchema):
    """Raw log frame produced by readers / fluentd."""

    FIELDS = {
        "__version__": (1, STRING),
        "logID": (2, STRING),
        "log": (3, STRING),
        "logSource": (4, STRING),
        "hostname": (5, STRING),
    }


class ParserSchema(BaseSchema):
    """Parsed log frame produced by parser stages."""

    FIELDS = {
        "__version__": (1, STRING),
        "parserType": (2, STRING),
        "parserID": (3, STRING),
        "EventID": (4, INT32),
        "template": (5, STRING),
        "variables": (6, REP_STRING),
        "parsedLogID": (7, STRING),
        "logID": (8, STRING),
        "log": (9, STRING),
        "logFormatVariables": (10, MAP_SS),
        "receivedTimestamp": (11, INT32),
        "parsedTimestamp": (12, INT32),
    }


class DetectorSchema(BaseSchema):
    """Anomaly alert frame produced by detector stages (field 7 unused)."""

    FIELDS = {
        "__version__": (1, STRING),
        "detectorID": (2, STRING),
        "detectorType": (3, STRING),
        "alertID": (4, STRING),
        "detectionTimestamp": (5, INT32),
        "logIDs": (6, REP_STRING),
        "score": (8, FLOAT),
        "extractedTimestamps": (9, REP_INT32),
        "description": (10, STRING),
        "receivedTimestamp": (11, INT32),
        "alertsObtain": (12, MAP_SS),
    }


class OutputSchema(BaseSchema):
    """Aggregated output frame at the pipeline egress."""

    FIELDS = {
        "__version__": (1, STRING),
        "detectorIDs": (2, REP_STRING),
        "detectorTypes": (3, REP_STRING),
        "alertIDs": (4, REP_STRING),
        "outputTimestamp": (5, INT32),
        "logIDs": (6, REP_STRING),
        "extractedTimestamps": (9, REP_INT32),
        "description": (10, STRING),
        "alertsObtain": (12, MAP_SS),
    }


__all__ = [
    "BaseSchema",
    "Schema",
    "LogSchema",
    "ParserSchema",
    "DetectorSchema",
    "OutputSchema",
    "SCHEMA_VERSION",
]
