"""ComponentResolver: short class name → full dotted path + config path.

Reference parity (/root/reference/src/service/features/component_resolver.py):
``resolve("NewValueDetector")`` walks the component library's packages with
``pkgutil.walk_packages``, matches a ``CoreComponent`` subclass by
``__name__`` (resolver.py:75-93), and locates ``<ClassName>Config`` in the
same module, falling back to the base ``CoreConfig``
(component_resolver.py:98-123).

The default library root here is this framework's first-class component
library ``detectmateservice_amd.library`` (the reference's external
``detectmatelibrary``); tests monkeypatch ``DEFAULT_ROOT`` exactly as the
reference's tests do (test_component_loader.py:21-46).
"""
from __future__ import annotations

import importlib
import inspect
import logging
import pkgutil
from typing import Optional, Tuple

from .base import CoreComponent, CoreConfig

DEFAULT_ROOT = "detectmateservice_amd.library"


class ComponentResolutionError(Exception):
    pass


class ComponentResolver:
    def __init__(self, root_package: str = DEFAULT_ROOT, logger: Optional[logging.Logger] = None) -> None:
        self.root_package = root_package
        self._log = logger or logging.getLogger(__name__)

    def resolve(self, component_type: str) -> Tuple[str, Optional[str]]:
        """Return (dotted component path, dotted config class path or None)."""
        if "." in component_type:
            # Already a dotted path; config found beside the class if possible.
            module_path, cls_name = component_type.rsplit(".", 1)
            cfg = self._find_config_in_module(module_path, cls_name)
            return component_type, cfg
        found = self._search_for_class(component_type)
        if found is None:
            raise ComponentResolutionError(
                f"component {component_type!r} not found under {self.root_package!r}"
            )
        module_path, cls_name = found
        cfg = self._find_config_in_module(module_path, cls_name)
        return f"{module_path}.{cls_name}", cfg

    def _search_for_class(self, cls_name: str) -> Optional[Tuple[str, str]]:
        try:
            root = importlib.import_module(self.root_package)
        except ImportError as exc:
            raise ComponentResolutionError(
                f"component library {self.root_package!r} is not importable: {exc}"
            ) from exc
        candidates = [self.root_package]
        if hasattr(root, "__path__"):
            for info in pkgutil.walk_packages(root.__path__, prefix=self.root_package + "."):
                candidates.append(info.name)
        for module_path in candidates:
            try:
                module = importlib.import_module(module_path)
            except Exception as exc:  # noqa: BLE001 - skip broken modules like the reference
                self._log.debug("skipping module %s: %s", module_path, exc)
                continue
            obj = getattr(module, cls_name, None)
            if (
                obj is not None
                and inspect.isclass(obj)
                and issubclass(obj, CoreComponent)
                and obj.__name__ == cls_name
            ):
                return module_path, cls_name
        return None

    def _find_config_in_module(self, module_path: str, cls_name: str) -> Optional[str]:
        """``<ClassName>Config`` beside the class, else None (base CoreConfig)."""
        try:
            module = importlib.import_module(module_path)
        except Exception:  # noqa: BLE001
            return None
        cfg_name = f"{cls_name}Config"
        obj = getattr(module, cfg_name, None)
        if obj is not None and inspect.isclass(obj) and issubclass(obj, CoreConfig):
            return f"{module_path}.{cfg_name}"
        return None
