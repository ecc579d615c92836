"""Elastic collectives: drop-don't-block semantics on the RCCL path.

SURVEY.md §7 names this the hard part of moving the reference's
elasticity onto collectives: RCCL assumes all ranks present, while the
reference's engine must survive a dead/absent peer (engine.py:281-301
there). The design here:

* every collective runs under a watchdog deadline; a peer death surfaces
  as a timeout/exception instead of an indefinite hang (process groups
  are created with an explicit timeout),
* on failure the wrapper counts a drop and returns the local-only result
  — the pipeline keeps processing, exactly like the reference's
  retry-then-drop socket sends,
* when the control plane reports the peer set healthy again,
  :func:`reform_group` tears down and re-initializes the process group
  with the new membership (communicator rebuild),
* while degraded, frames that would have moved over xGMI can be diverted
  to the socket transport (the Service path) — the cold-path fallback.
"""
from __future__ import annotations

import datetime
import logging
import os
from typing import Callable, List, Optional, Sequence

import torch
import torch.distributed as dist

log = logging.getLogger(__name__)


class ElasticComm:
    """Wraps collectives with failure accounting and local fallbacks."""

    def __init__(self, timeout_s: float = 30.0) -> None:
        self.timeout_s = timeout_s
        self.drops = 0
        self.degraded = False

    # -- group lifecycle ----------------------------------------------
    def init_from_env(self, backend: Optional[str] = None) -> tuple:
        world = int(os.environ.get("WORLD_SIZE", "1"))
        rank = int(os.environ.get("RANK", "0"))
        if world > 1 and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if torch.cuda.is_available() else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            dist.init_process_group(
                backend=backend, rank=rank, world_size=world,
                timeout=datetime.timedelta(seconds=self.timeout_s),
            )
        return rank, world

    def reform_group(
        self, rank: int, world_size: int, backend: Optional[str] = None,
        master_port: Optional[int] = None,
    ) -> None:
        """Communicator rebuild after membership change: destroy + re-init
        with the surviving rank set (ranks must be renumbered contiguously
        by the control plane before calling)."""
        if dist.is_initialized():
            try:
                dist.destroy_process_group()
            except Exception as exc:  # noqa: BLE001
                log.warning("destroy_process_group during reform: %s", exc)
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if master_port is not None:
            os.environ["MASTER_PORT"] = str(master_port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world_size,
            timeout=datetime.timedelta(seconds=self.timeout_s),
        )
        self.degraded = False
        log.info("process group reformed: rank %d / world %d", rank, world_size)

    # -- guarded collectives -------------------------------------------
    def _guard(self, op: Callable, fallback):
        if self.degraded or not dist.is_initialized() or dist.get_world_size() == 1:
            return fallback() if callable(fallback) else fallback
        try:
            return op()
        except Exception as exc:  # noqa: BLE001 - RCCL/gloo raise varied types
            self.drops += 1
            self.degraded = True
            log.error("collective failed (%s); degrading to local-only", exc)
            return fallback() if callable(fallback) else fallback

    def all_gather_summaries(self, summary: torch.Tensor) -> torch.Tensor:
        def op():
            out = [torch.empty_like(summary) for _ in range(dist.get_world_size())]
            dist.all_gather(out, summary)
            return torch.stack(out)

        return self._guard(op, lambda: summary.unsqueeze(0))

    def broadcast_packed(self, lines, lens, src: int, device) -> tuple:
        from . import dist as dmx_dist

        def op():
            return dmx_dist.broadcast_packed(lines, lens, src, device)

        def fallback():
            if lines is None:
                raise RuntimeError(
                    "broadcast fallback on a sink rank has no local data; "
                    "divert this stage to the socket transport"
                )
            return lines, lens

        return self._guard(op, fallback)

    def send_packed(self, lines, lens, dst: int) -> bool:
        from . import dist as dmx_dist

        def op():
            dmx_dist.send_packed(lines, lens, dst)
            return True

        return self._guard(op, False)
