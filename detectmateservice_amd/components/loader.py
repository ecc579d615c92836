"""ComponentLoader + ConfigClassLoader.

Reference parity:

* ``ComponentLoader.load_component`` (component_loader.py:13-67): import
  ``module.ClassName`` — try the path as-is, then prefixed with the library
  root (component_loader.py:34-43); instantiate with ``config=`` kwarg only
  when config is truthy (47-50); enforce ``isinstance(instance,
  CoreComponent)`` (52-55).
* ``ConfigClassLoader.load_config_class`` (config_loader.py:16-80):
  import a config class by dotted path with library-relative then absolute
  fallback (48-63); enforce ``issubclass(cls, CoreConfig)`` (68-69).
"""
from __future__ import annotations

import importlib
import logging
from typing import Any, Optional, Type

from .base import CoreComponent, CoreConfig

DEFAULT_ROOT = "detectmateservice_amd.library"
BASE_PACKAGE = "detectmateservice_amd.library"


def normalize_component_config(config):
    """Library-side config pipeline (reference docs/interfaces.md:74-82):
    flatten ``params`` into the top level (without removing the original
    block) and strip the ``all_`` prefix from keys (an ``all_x`` key
    applies ``x`` to every instance)."""
    if not isinstance(config, dict):
        return config
    out = dict(config)
    params = out.get("params")
    if isinstance(params, dict):
        for k, v in params.items():
            out.setdefault(k, v)
    for k in list(out.keys()):
        if isinstance(k, str) and k.startswith("all_"):
            out.setdefault(k[4:], out[k])
    return out


class ComponentLoadError(Exception):
    pass


class ComponentLoader:
    def __init__(self, root_package: str = DEFAULT_ROOT, logger: Optional[logging.Logger] = None) -> None:
        self.root_package = root_package
        self._log = logger or logging.getLogger(__name__)

    def load_component(
        self,
        component_path: str,
        config: Any = None,
        logger: Optional[logging.Logger] = None,
    ) -> CoreComponent:
        module_path, cls_name = component_path.rsplit(".", 1)
        module = None
        errors = []
        for candidate in (module_path, f"{self.root_package}.{module_path}"):
            try:
                module = importlib.import_module(candidate)
                break
            except ImportError as exc:
                errors.append(f"{candidate}: {exc}")
        if module is None:
            raise ComponentLoadError(
                f"cannot import component module for {component_path!r}: {'; '.join(errors)}"
            )
        cls = getattr(module, cls_name, None)
        if cls is None:
            raise ComponentLoadError(
                f"module {module.__name__!r} has no class {cls_name!r}"
            )
        config = normalize_component_config(config)
        try:
            instance = cls(config=config) if config else cls()
        except TypeError as exc:
            raise ComponentLoadError(
                f"could not instantiate {component_path!r}: {exc}"
            ) from exc
        if not isinstance(instance, CoreComponent):
            raise ComponentLoadError(
                f"{component_path!r} is not a CoreComponent (got {type(instance)!r})"
            )
        return instance


class ConfigClassLoader:
    def __init__(self, base_package: str = BASE_PACKAGE, logger: Optional[logging.Logger] = None) -> None:
        self.base_package = base_package
        self._log = logger or logging.getLogger(__name__)

    def load_config_class(self, config_path: str) -> Type[CoreConfig]:
        module_path, cls_name = config_path.rsplit(".", 1)
        module = None
        errors = []
        for candidate in (f"{self.base_package}.{module_path}", module_path):
            try:
                module = importlib.import_module(candidate)
                if getattr(module, cls_name, None) is not None:
                    break
            except ImportError as exc:
                errors.append(f"{candidate}: {exc}")
                module = None
        if module is None:
            raise ComponentLoadError(
                f"cannot import config module for {config_path!r}: {'; '.join(errors)}"
            )
        cls = getattr(module, cls_name, None)
        if cls is None:
            raise ComponentLoadError(f"module {module.__name__!r} has no class {cls_name!r}")
        if not (isinstance(cls, type) and issubclass(cls, CoreConfig)):
            raise ComponentLoadError(f"{config_path!r} is not a CoreConfig subclass")
        return cls
