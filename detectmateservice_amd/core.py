"""Service: lifecycle orchestration for one pipeline component.

Reference parity (/root/reference/src/service/core.py:64-436): the Service
resolves the component type (core.py:91), builds a ConfigManager (121),
loads the library component (145), initializes the Engine with itself as
processor (155), starts the WebServer + engine in ``run()`` (214-229) and
parks the main thread on an exit event (229). ``process()`` counts
bytes/lines and times the component call (184-206); ``status()`` returns
``{status, settings, configs}`` (386-423); ``reconfigure`` validates and
swaps the ConfigManager state and optionally persists with defaults
stripped (299-338); ``shutdown`` wakes ``run()`` which stops the web
server and the engine (347-353, 229-236). Context-manager protocol calls
``setup_io()`` on enter (424-436, 209).

MI355X-first extension: ``process_batch`` is the native path — the engine
hands the whole drained batch to the component in one call (GPU kernels
run per batch); ``process`` remains the compatible per-message API.
"""
from __future__ import annotations

import logging
import threading
import time
from pathlib import Path
from typing import Any, Dict, List, Optional

from .components.base import CoreComponent
from .components.config_manager import ConfigManager
from .components.loader import ComponentLoader, ConfigClassLoader
from .components.resolver import ComponentResolver
from .engine.engine import Engine
from .metadata import __version__
from .settings import ServiceSettings
from .utils.logging import build_service_logger
from .utils.metrics import ServiceMetrics


class Service:
    """One OS process = one Service wrapping one pipeline component."""

    def __init__(
        self,
        settings: ServiceSettings,
        component_config: Optional[Dict[str, Any]] = None,
        socket_factory=None,
    ) -> None:
        # --- settings-driven distributed placement (dist_mode) ---------
        # Initialize the process group from torchrun env vars, then
        # substitute {rank}/{world} placeholders in addresses/identity so
        # ONE YAML launched as `torchrun --nproc-per-node N detectmate
        # --settings x.yaml` places N ranks (reference principle: the
        # topology lives in config, parser_settings.yaml out_addr).
        self.dist_ctx = None
        if settings.dist_mode is not None:
            from types import SimpleNamespace

            from .parallel import dist as dmx_dist

            rank, world = dmx_dist.init_from_env(
                settings.dist_backend, timeout_s=settings.dist_timeout_s)
            settings = settings.resolve_dist_placeholders(rank, world)
            if world > 1:
                self.dist_ctx = SimpleNamespace(
                    mode=settings.dist_mode, rank=rank, world=world,
                    src=settings.dist_src_rank,
                )
        self.settings = settings
        self.logger = build_service_logger(
            settings.component_type,
            settings.component_id or "unknown",
            settings.log_level,
            settings.log_dir,
        )
        self.metrics = ServiceMetrics(
            settings.component_type, settings.component_id or "unknown"
        )
        self._service_exit_event = threading.Event()
        self._running = False

        # --- component resolution + loading (reference core.py:87-152) ---
        self.library_component: Optional[CoreComponent] = None
        self._component_class_name: Optional[str] = None
        self._config_class_path: Optional[str] = settings.config_class
        component_path: Optional[str] = None
        if settings.component_type and not settings.component_type.startswith("core"):
            resolver = ComponentResolver(logger=self.logger)
            component_path, resolved_cfg = resolver.resolve(settings.component_type)
            self._component_class_name = component_path.rsplit(".", 1)[1]
            if self._config_class_path is None:
                self._config_class_path = resolved_cfg

        schema = None
        if self._config_class_path:
            try:
                schema = ConfigClassLoader(logger=self.logger).load_config_class(
                    self._config_class_path
                )
            except Exception as exc:  # noqa: BLE001
                self.logger.warning(
                    "could not load config class %s: %s", self._config_class_path, exc
                )
        self.config_manager = ConfigManager(
            settings.config_file, schema=schema, logger=self.logger
        )

        self._component_path = component_path
        if component_path is not None:
            cfg = component_config
            if cfg is None and self._component_class_name:
                cfg = self.config_manager.component_section(self._component_class_name)
            self.library_component = ComponentLoader(logger=self.logger).load_component(
                component_path, config=cfg, logger=self.logger
            )

        # source-mode sanity: fail at construction, not in the engine thread
        if settings.engine_source_mode and (
            self.library_component is None
            or not hasattr(self.library_component, "stream_batches")
        ):
            raise ValueError(
                f"engine_source_mode requires a component with stream_batches "
                f"(a reader); {settings.component_type!r} has none"
            )

        # --- engine (reference core.py:155) ---
        self.engine = Engine(
            settings,
            processor=self,
            socket_factory=socket_factory,
            logger=self.logger,
            metrics=self.metrics,
            dist_ctx=self.dist_ctx,
        )

        # --- admin web server (constructed unconditionally, core.py:81) ---
        from .web.server import WebServer

        self.web_server = WebServer(self) if settings.http_enabled else None

    # ------------------------------------------------------------------
    # processing (reference core.py:176-206)
    # ------------------------------------------------------------------
    def process(self, data: bytes) -> Optional[bytes]:
        out = self.process_batch([data])
        return out[0]

    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]:
        if self.library_component is not None:
            return self.library_component.process_batch(frames)
        # passthrough mode for core* component types (reference core.py:204-206)
        return list(frames)

    # -- packed data plane (engine_packed_mode) ------------------------
    def supports_packed_frames(self) -> bool:
        return hasattr(self.library_component, "process_packed_frames")

    def packed_max_len(self) -> int:
        fn = getattr(self.library_component, "packed_max_len", None)
        return fn() if fn is not None else 256

    def packed_pin_memory(self) -> bool:
        fn = getattr(self.library_component, "packed_pin_memory", None)
        return fn() if fn is not None else False

    def process_packed_frames(self, lines, lens, ids_blob, ids_off):
        return self.library_component.process_packed_frames(
            lines, lens, ids_blob, ids_off
        )

    @property
    def submit_packed_frames(self):
        return getattr(self.library_component, "submit_packed_frames", None)

    @property
    def collect_packed_frames(self):
        return getattr(self.library_component, "collect_packed_frames", None)

    def source_batches(self, batch_size: int, stop_event):
        """Source-mode delegation (engine_source_mode: reader services)."""
        if self.library_component is None or not hasattr(
            self.library_component, "stream_batches"
        ):
            raise RuntimeError(
                f"component {self.settings.component_type!r} does not support "
                "source mode (no stream_batches)"
            )
        return self.library_component.stream_batches(batch_size, stop_event)

    # ------------------------------------------------------------------
    # lifecycle (reference core.py:209-353)
    # ------------------------------------------------------------------
    def setup_io(self) -> None:
        """Model-loading hook (reference core.py:209): delegates to the
        component's setup (GPU weight upload happens here)."""
        if self.library_component is not None:
            self.library_component.setup()

    def run(self) -> None:
        """Blocking main loop: start admin + engine, wait for shutdown."""
        if self.web_server is not None:
            self.web_server.start()
        if self.settings.engine_autostart:
            self.start()
        self._service_exit_event.wait()
        if self.web_server is not None:
            self.web_server.stop()
        try:
            self.stop()
        except Exception as exc:  # noqa: BLE001
            self.logger.error("engine stop during shutdown failed: %s", exc)
        self.engine.close()
        if self.library_component is not None:
            self.library_component.teardown()

    def start(self) -> None:
        self.engine.start()
        self._running = True
        self.logger.info("service started (v%s)", __version__)

    def stop(self) -> None:
        self.engine.stop()
        self._running = False
        self.logger.info("service stopped")

    def status(self) -> Dict[str, Any]:
        return self._create_status_report(self._running)

    def reconfigure(
        self, config: Dict[str, Any], persist: bool = False, reload: bool = False
    ) -> Dict[str, Any]:
        """Validate + swap config; optionally persist defaults-stripped YAML
        (reference core.py:299-338).

        ``reload`` (no reference equivalent — the reference's live
        component keeps its constructor-time config, core.py:299-345
        comment): rebuild the library component from the NEW config and
        swap it in atomically. In-memory detector state (learned values,
        calibration) restarts fresh; checkpoint/restore first to keep it.
        """
        validated = self.config_manager.update(config)
        if persist:
            self.config_manager.save(self.config_manager.get())
        if reload and self._component_path is not None:
            cfg = None
            if self._component_class_name:
                cfg = self.config_manager.component_section(self._component_class_name)
            new_component = ComponentLoader(logger=self.logger).load_component(
                self._component_path, config=cfg, logger=self.logger
            )
            old, self.library_component = self.library_component, new_component
            if old is not None:
                old.teardown()
            self.logger.info("component reloaded from new config")
        self.logger.info("reconfigured (persist=%s reload=%s)", persist, reload)
        return self.config_manager.get()

    def shutdown(self) -> None:
        self._service_exit_event.set()

    # ------------------------------------------------------------------
    def rescore(self, lines_back: int,
                threshold: Optional[float] = None) -> Dict[str, Any]:
        """Post-hoc re-score of the component's HBM line-buffer window
        (BASELINE config 5); refuses when the component has no buffer."""
        fn = getattr(self.library_component, "rescore_window", None)
        if fn is None:
            return {"rescored": 0,
                    "reason": f"component {self.settings.component_type!r} "
                              "has no line buffer"}
        return fn(lines_back, threshold)

    # ------------------------------------------------------------------
    def dp_sync(self) -> Dict[str, Any]:
        """Merge data-parallel detector state across ranks (dist_mode
        "dp"): a COLLECTIVE — every rank must call it (operators hit
        POST /admin/dp-sync on all ranks, or a cadence task does).
        Components implement ``dist_sync()``; NewValue detectors merge
        learned value sets, the fused GPU detector all-reduces its hash
        tables (parallel/dist.py::all_reduce_hashsets)."""
        import torch.distributed as tdist

        if self.dist_ctx is None or not tdist.is_initialized():
            return {"synced": False, "reason": "no process group"}
        fn = getattr(self.library_component, "dist_sync", None)
        if fn is None:
            return {"synced": False,
                    "reason": f"component {self.settings.component_type!r} "
                              "has no dist_sync"}
        fn()
        self.logger.info("dp_sync complete (world=%d)", self.dist_ctx.world)
        return {"synced": True, "world": self.dist_ctx.world}

    # ------------------------------------------------------------------
    # checkpoint / resume (the reference has none; this
    # framework's detector state — GPU hash sets, transformer weights,
    # calibration — must survive restarts)
    # ------------------------------------------------------------------
    def _resolve_checkpoint_path(self, path: str | Path, from_admin: bool) -> Path:
        """Admin calls (unauthenticated HTTP) are confined to the
        configured ``checkpoint_dir``; direct Python calls by the
        operator's own process keep arbitrary paths."""
        p = Path(path)
        if not from_admin:
            return p
        ckdir = self.settings.checkpoint_dir
        if ckdir is None:
            raise PermissionError(
                "admin checkpoint/restore requires settings.checkpoint_dir"
            )
        base = Path(ckdir).resolve()
        cand = (p if p.is_absolute() else base / p).resolve()
        if not cand.is_relative_to(base):
            raise PermissionError(
                f"checkpoint path {str(path)!r} escapes checkpoint_dir"
            )
        return cand

    def checkpoint(self, path: str | Path, from_admin: bool = False) -> Dict[str, Any]:
        import torch

        path = self._resolve_checkpoint_path(path, from_admin)
        path.parent.mkdir(parents=True, exist_ok=True)
        state = {
            "version": __version__,
            "component_type": self.settings.component_type,
            "component_id": self.settings.component_id,
            "component_state": (
                self.library_component.state_dict()
                if self.library_component is not None
                else {}
            ),
            "config": self.config_manager.get(),
            "timestamp": time.time(),
        }
        tmp = path.with_suffix(path.suffix + ".tmp")
        torch.save(state, tmp)
        tmp.rename(path)  # atomic publish
        self.logger.info("checkpoint written to %s", path)
        return {"path": str(path), "timestamp": state["timestamp"]}

    def restore(self, path: str | Path, from_admin: bool = False) -> Dict[str, Any]:
        import torch

        path = self._resolve_checkpoint_path(path, from_admin)
        # weights_only: checkpoints are tensors + primitives only — a
        # planted pickle payload must raise here, not execute
        # (component state_dicts keep to that contract; sklearn models
        # store their pickle as inert bytes unpickled only by their own
        # load_state_dict after this trust gate).
        state = torch.load(path, map_location="cpu", weights_only=True)
        if state.get("component_type") != self.settings.component_type:
            raise ValueError(
                f"checkpoint is for component_type={state.get('component_type')!r}, "
                f"this service is {self.settings.component_type!r}"
            )
        if self.library_component is not None and state.get("component_state"):
            self.library_component.load_state_dict(state["component_state"])
        self.logger.info("restored component state from %s", path)
        return {"path": str(path), "timestamp": state.get("timestamp")}

    # ------------------------------------------------------------------
    def _create_status_report(self, running: bool) -> Dict[str, Any]:
        m = self.metrics

        def _val(c):
            try:
                return c._value.get()
            except AttributeError:
                return None

        return {
            "status": {
                "running": running,
                "engine_running": self.engine.running,
                "version": __version__,
                "timestamp": int(time.time()),
            },
            "metrics": {
                "read_lines": _val(m.data_read_lines_total),
                "processed_lines": _val(m.data_processed_lines_total),
                "written_lines": _val(m.data_written_lines_total),
                "dropped_lines": _val(m.data_dropped_lines_total),
                "processing_errors": _val(m.processing_errors_total),
            },
            "settings": self.settings.model_dump(mode="json"),
            "configs": self.config_manager.get(),
        }

    # -- context manager (reference core.py:424-436) -------------------
    def __enter__(self) -> "Service":
        self.setup_io()
        return self

    def __exit__(self, exc_type, exc, tb) -> None:
        self.shutdown()
        # Give run() a moment to unwind if it is executing in this thread's
        # caller; direct cleanup otherwise.
        if not self._service_exit_event.is_set():
            self._service_exit_event.set()
