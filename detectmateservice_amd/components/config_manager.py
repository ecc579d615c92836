"""ConfigManager: load / validate / persist the component config YAML.

Reference parity (/root/reference/src/service/features/config_manager.py):

* the component config file has the nested ``detectors:/parsers:/readers:``
  shape and is validated against that wrapper, not the component schema
  (config_manager.py:12-15, comment 53-60),
* RLock-guarded state (28),
* a default file is materialized from the schema when missing (37-44),
* ``save()`` prefers ``to_dict()`` (defaults stripped) over ``model_dump``
  (85-92),
* ``update()`` swaps the validated in-memory copy (118-123); the running
  component keeps its constructor-time config — reconfigure is "visible at
  status/next load" (reference semantics).
"""
from __future__ import annotations

import logging
import threading
from pathlib import Path
from typing import Any, Dict, Optional, Type

import yaml
from pydantic import BaseModel, ConfigDict

from .base import CoreConfig


class ServiceConfig(BaseModel):
    """Nested component-config wrapper (reference config_manager.py:12-15)."""

    model_config = ConfigDict(extra="allow")
    detectors: Dict[str, Any] = {}
    parsers: Dict[str, Any] = {}
    readers: Dict[str, Any] = {}


class ConfigManagerError(Exception):
    pass


class ConfigManager:
    def __init__(
        self,
        config_file: Optional[Path],
        schema: Optional[Type[CoreConfig]] = None,
        logger: Optional[logging.Logger] = None,
    ) -> None:
        self.config_file = Path(config_file) if config_file else None
        self.schema = schema
        self._log = logger or logging.getLogger(__name__)
        self._lock = threading.RLock()
        self._config: ServiceConfig = ServiceConfig()
        if self.config_file is not None:
            self.load()

    def load(self) -> ServiceConfig:
        with self._lock:
            if self.config_file is None:
                return self._config
            if not self.config_file.exists():
                # Materialize a default config file from the schema
                # (reference config_manager.py:37-44).
                self._config = ServiceConfig()
                self.save(self._config.model_dump(exclude_defaults=True))
                return self._config
            with open(self.config_file, "r", encoding="utf-8") as fh:
                data = yaml.safe_load(fh) or {}
            if not isinstance(data, dict):
                raise ConfigManagerError(
                    f"component config {self.config_file} must be a YAML mapping"
                )
            self._config = ServiceConfig.model_validate(data)
            return self._config

    def save(self, data: Optional[Dict[str, Any]] = None) -> None:
        with self._lock:
            if self.config_file is None:
                raise ConfigManagerError("no config_file configured; cannot persist")
            if data is None:
                data = self._to_plain(self._config)
            self.config_file.parent.mkdir(parents=True, exist_ok=True)
            with open(self.config_file, "w", encoding="utf-8") as fh:
                yaml.safe_dump(data, fh, default_flow_style=False, sort_keys=False)
            self._log.debug("persisted component config to %s", self.config_file)

    def update(self, data: Dict[str, Any]) -> ServiceConfig:
        with self._lock:
            self._config = ServiceConfig.model_validate(data)
            return self._config

    def get(self) -> Dict[str, Any]:
        with self._lock:
            return self._to_plain(self._config)

    @staticmethod
    def _to_plain(cfg: ServiceConfig) -> Dict[str, Any]:
        """Defaults-stripped dict (reference's to_dict preference, 85-92)."""
        if hasattr(cfg, "to_dict"):
            try:
                return cfg.to_dict()  # type: ignore[attr-defined]
            except TypeError:
                pass
        return cfg.model_dump(exclude_defaults=True)

    def component_section(self, class_name: str) -> Optional[Dict[str, Any]]:
        """Extract the config block for one component class from the nested
        detectors:/parsers:/readers: structure."""
        with self._lock:
            for section in ("detectors", "parsers", "readers"):
                block = getattr(self._config, section, {}) or {}
                if class_name in block:
                    return block[class_name]
        return None
