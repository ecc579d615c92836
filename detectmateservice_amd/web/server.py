"""Admin web server: FastAPI + uvicorn in a daemon thread.

Reference parity (/root/reference/src/service/features/web/server.py:14-48):
the Service is injected through ``dependency_overrides[get_service]``; the
``/metrics`` endpoint returns ``prometheus_client.generate_latest()``;
uvicorn runs without signal handlers because it lives off the main thread.
"""
from __future__ import annotations

import threading
import time
from typing import TYPE_CHECKING

import uvicorn
from fastapi import FastAPI
from fastapi.responses import PlainTextResponse
from prometheus_client import generate_latest

from .router import get_service, router

if TYPE_CHECKING:  # pragma: no cover
    from ..core import Service


def build_app(service: "Service") -> FastAPI:
    app = FastAPI(title="detectmate-mi355x admin", docs_url=None, redoc_url=None)
    app.include_router(router)
    app.dependency_overrides[get_service] = lambda: service

    @app.get("/metrics", response_class=PlainTextResponse)
    def metrics() -> str:
        return generate_latest().decode("utf-8")

    return app


class WebServer(threading.Thread):
    def __init__(self, service: "Service") -> None:
        super().__init__(name="WebServerThread", daemon=True)
        self.service = service
        self.app = build_app(service)
        config = uvicorn.Config(
            self.app,
            host=service.settings.http_host,
            port=service.settings.http_port,
            log_level="warning",
            # no signal handlers off the main thread (reference server.py:40-42)
        )
        self._server = uvicorn.Server(config)
        self._server.install_signal_handlers = lambda: None  # type: ignore[assignment]

    def run(self) -> None:
        self._server.run()

    def stop(self) -> None:
        self._server.should_exit = True
        self.join(timeout=3.0)
        if self.is_alive():
            # keep-alive connections can hold graceful shutdown open;
            # force-close them (uvicorn's force_exit path)
            self._server.force_exit = True
            self.join(timeout=3.0)

    def wait_started(self, timeout_s: float = 10.0) -> bool:
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline:
            if getattr(self._server, "started", False):
                return True
            time.sleep(0.02)
        return False
