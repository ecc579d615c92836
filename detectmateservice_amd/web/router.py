"""Admin HTTP router.

Reference parity (/root/reference/src/service/features/web/router.py:18-46):
``POST /admin/start|stop|reconfigure|shutdown``, ``GET /admin/status``.
The Service is injected via a FastAPI dependency override
(web/server.py:31 in the reference).
"""
from __future__ import annotations

from typing import Any, Dict, Optional

from fastapi import APIRouter, Depends, HTTPException
from pydantic import BaseModel

router = APIRouter(prefix="/admin")


def get_service():  # overridden by WebServer with the live Service
    raise RuntimeError("service dependency not wired")


class ReconfigPayload(BaseModel):
    config: Dict[str, Any]
    persist: bool = False
    reload: bool = False  # rebuild the live component from the new config


@router.post("/start")
def start(service=Depends(get_service)) -> Dict[str, Any]:
    service.start()
    return {"status": "started"}


@router.post("/stop")
def stop(service=Depends(get_service)) -> Dict[str, Any]:
    service.stop()
    return {"status": "stopped"}


@router.get("/status")
def status(service=Depends(get_service)) -> Dict[str, Any]:
    return service.status()


@router.post("/reconfigure")
def reconfigure(payload: ReconfigPayload, service=Depends(get_service)) -> Dict[str, Any]:
    configs = service.reconfigure(payload.config, payload.persist, payload.reload)
    return {"status": "reconfigured", "configs": configs}


@router.post("/shutdown")
def shutdown(service=Depends(get_service)) -> Dict[str, Any]:
    service.shutdown()
    return {"status": "shutting down"}


class RescorePayload(BaseModel):
    lines_back: int
    threshold: Optional[float] = None


@router.post("/rescore")
def rescore(payload: RescorePayload, service=Depends(get_service)) -> Dict[str, Any]:
    """Re-score the newest N resident lines of the HBM line buffer
    (config 5) with an optional alternative threshold."""
    return service.rescore(payload.lines_back, payload.threshold)


@router.post("/dp-sync")
def dp_sync(service=Depends(get_service)) -> Dict[str, Any]:
    """Collective merge of data-parallel detector state (dist_mode
    "dp"); call on EVERY rank."""
    return service.dp_sync()


class CheckpointPayload(BaseModel):
    #: name relative to settings.checkpoint_dir (absolute paths must
    #: resolve inside it); refused with 403 when no checkpoint_dir is set
    path: str


@router.post("/checkpoint")
def checkpoint(payload: CheckpointPayload, service=Depends(get_service)) -> Dict[str, Any]:
    try:
        return service.checkpoint(payload.path, from_admin=True)
    except PermissionError as exc:
        raise HTTPException(status_code=403, detail=str(exc)) from None


@router.post("/restore")
def restore(payload: CheckpointPayload, service=Depends(get_service)) -> Dict[str, Any]:
    try:
        return service.restore(payload.path, from_admin=True)
    except PermissionError as exc:
        raise HTTPException(status_code=403, detail=str(exc)) from None
