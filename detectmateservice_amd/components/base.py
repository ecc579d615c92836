"""Component contract: what the reference delegates to ``detectmatelibrary``.

The reference loads components from an external package and requires only
``CoreComponent.process(bytes) -> bytes | None`` plus a pydantic
``CoreConfig`` (spec: /root/reference/docs/interfaces.md:5-57; enforcement:
component_loader.py:52-55, config_loader.py:68-69). This framework provides
the contract natively, with one MI355X-first extension: ``process_batch``
— the engine drains N frames and hands the whole batch to the component so
stage compute (parse kernels, detector scoring, MFMA GEMMs) runs batched
on the GPU. The default ``process_batch`` falls back to per-frame
``process`` so reference-shaped components work unmodified.
"""
from __future__ import annotations

import enum
from abc import ABC, abstractmethod
from typing import Any, Dict, List, Optional, Type, Union

from pydantic import BaseModel, ConfigDict


class CoreConfig(BaseModel):
    """Base component config (reference docs/interfaces.md:49-57).

    Pydantic with ``model_validate`` / ``model_dump``; ``to_dict()`` is the
    no-defaults serialization the reference prefers when persisting
    (core.py:326-327, config_manager.py:85-92).
    """

    model_config = ConfigDict(extra="allow")

    def to_dict(self) -> Dict[str, Any]:
        return self.model_dump(exclude_defaults=True, exclude_none=True)


class CoreComponent(ABC):
    """Base pipeline component (reader / parser / detector).

    ``__init__(config: dict | CoreConfig | None)`` and
    ``process(bytes) -> bytes | None`` per docs/interfaces.md:12-35.
    Returning ``None`` means "filtered" — the engine sends nothing
    downstream (reference engine.py:238-240).
    """

    CONFIG_CLASS: Type[CoreConfig] = CoreConfig

    def __init__(self, config: Union[Dict[str, Any], CoreConfig, None] = None) -> None:
        if config is None:
            self.config = self.CONFIG_CLASS()
        elif isinstance(config, CoreConfig):
            self.config = config
        elif isinstance(config, dict):
            self.config = self.CONFIG_CLASS.model_validate(config)
        else:
            raise TypeError(f"config must be dict/CoreConfig/None, got {type(config)}")
        # method_type check (reference library config pipeline,
        # docs/interfaces.md:74-82 "check_type"): a config naming a
        # different method_type than the component's declared default is a
        # wiring error, not a silent override.
        declared = self.CONFIG_CLASS.model_fields.get("method_type")
        if declared is not None and declared.default:
            given = getattr(self.config, "method_type", None)
            if given and given != declared.default:
                raise ValueError(
                    f"config method_type {given!r} does not match "
                    f"{type(self).__name__}'s {declared.default!r}"
                )

    @abstractmethod
    def process(self, data: bytes) -> Optional[bytes]:
        """Process one serialized frame; None filters it out."""

    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]:
        """Process a batch of frames (MI355X-native fast path).

        Components with GPU kernels override this; the default preserves
        per-frame semantics for reference-shaped components.
        """
        return [self.process(f) for f in frames]

    # -- optional lifecycle hooks --------------------------------------
    def setup(self) -> None:
        """Called once before the engine starts (model loading hook —
        mirrors the reference's ``Service.setup_io()`` seam, core.py:209)."""

    def teardown(self) -> None:
        """Called on service shutdown."""

    # -- state checkpointing (absent in the reference,
    #    required here because detector state may be GPU-resident) -------
    def state_dict(self) -> Dict[str, Any]:
        return {}

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        pass


class BufferMode(enum.Enum):
    """Detector buffering modes (reference docs/interfaces.md:167)."""

    NO_BUF = "no_buf"
    FIXED = "fixed"
    SLIDING = "sliding"


class CoreDetectorConfig(CoreConfig):
    """Detector base config (reference docs/interfaces.md:139-167).

    ``data_use_training: N`` = train on the first N lines then switch to
    detect (reference docs/getting_started.md:426-435). ``buffer_mode``
    (interfaces.md:167): NO_BUF detects per line; FIXED collects
    ``buffer_size`` lines and detects once per full window; SLIDING keeps
    the last ``buffer_size`` lines as context and detects per line once
    the window is primed.
    """

    data_use_training: int = 0
    buffer_mode: BufferMode = BufferMode.NO_BUF
    buffer_size: int = 16


class CoreDetector(CoreComponent):
    """Detector base: streaming train-then-detect over ParserSchema frames.

    Subclasses implement ``train(parsed_batch)`` and
    ``detect(parsed, alert) -> bool`` (reference docs/interfaces.md:139-167);
    the provided ``process``/``process_batch`` handle deserialization, the
    training-count switch and alert serialization.
    """

    CONFIG_CLASS: Type[CoreConfig] = CoreDetectorConfig

    def __init__(self, config: Union[Dict[str, Any], CoreConfig, None] = None) -> None:
        super().__init__(config)
        self._seen_lines = 0
        self._buffer: List[Any] = []  # FIXED/SLIDING window of ParserSchema

    # subclass API ------------------------------------------------------
    def train(self, parsed_batch: List[Any]) -> None:  # List[ParserSchema]
        raise NotImplementedError

    def detect(self, parsed: Any, alert: Any) -> bool:  # (ParserSchema, DetectorSchema)
        raise NotImplementedError

    def detect_window(self, window: List[Any], alert: Any) -> bool:
        """Windowed detection hook (FIXED/SLIDING buffer modes). Default:
        per-line detect on the newest window entry."""
        return self.detect(window[-1], alert)

    # framework-provided plumbing --------------------------------------
    def process(self, data: bytes) -> Optional[bytes]:
        out = self.process_batch([data])
        return out[0]

    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]:
        from ..schemas import DetectorSchema, ParserSchema

        parsed = [ParserSchema.deserialize(f) for f in frames]
        results: List[Optional[bytes]] = [None] * len(frames)
        n_train = int(getattr(self.config, "data_use_training", 0))

        train_upto = 0
        if self._seen_lines < n_train:
            train_upto = min(len(parsed), n_train - self._seen_lines)
            self.train(parsed[:train_upto])
        self._seen_lines += len(parsed)

        mode = getattr(self.config, "buffer_mode", BufferMode.NO_BUF)
        if isinstance(mode, str):
            mode = BufferMode(mode)
        size = max(1, int(getattr(self.config, "buffer_size", 16)))

        for i in range(train_upto, len(parsed)):
            alert = DetectorSchema()
            if mode == BufferMode.NO_BUF:
                if self.detect(parsed[i], alert):
                    results[i] = alert.serialize()
            elif mode == BufferMode.FIXED:
                self._buffer.append(parsed[i])
                if len(self._buffer) >= size:
                    window, self._buffer = self._buffer, []
                    if self.detect_window(window, alert):
                        results[i] = alert.serialize()
            else:  # SLIDING
                self._buffer.append(parsed[i])
                if len(self._buffer) > size:
                    self._buffer.pop(0)
                if len(self._buffer) == size and self.detect_window(
                    list(self._buffer), alert
                ):
                    results[i] = alert.serialize()
        return results
