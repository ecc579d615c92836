"""Distributed helpers: RCCL over xGMI intra-node, gloo on CPU.

MI355X mapping of the reference's process-level parallelism (SURVEY.md
§2.5): within a node, pipeline hops and fan-out move as device tensors
over torch.distributed (backend "nccl" IS RCCL on ROCm) instead of
per-process NNG sockets:

* pipeline hop (reader→parser→detector): point-to-point send/recv — one
  hop = one xGMI link, no ring (SURVEY.md §5.8 guidance),
* multi_output fan-out (1→N): broadcast from the producing stage's rank,
* data-parallel detectors: all_gather of per-rank anomaly summaries.

Edge transports (fluentd interop, cross-node) stay on the framed socket
layer in engine/sockets.py; these helpers are the intra-node fast path.
"""
from __future__ import annotations

import os
from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist


def init_from_env(backend: Optional[str] = None) -> Tuple[int, int]:
    """Initialize the default process group from torchrun env vars.

    Returns (rank, world_size); no-op (0, 1) when WORLD_SIZE is unset.
    Backend defaults to nccl (=RCCL) when a GPU is visible, else gloo.
    """
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return 0, 1
    rank = int(os.environ.get("RANK", "0"))
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
    return rank, world_size


# ---------------------------------------------------------------------------
# packed-batch transport: (lines u8 [B, max_len], lens i32 [B])
# ---------------------------------------------------------------------------

_HDR_LEN = 3  # [B, max_len, flags]


def send_packed(lines: torch.Tensor, lens: torch.Tensor, dst: int, tag: int = 0) -> None:
    """P2P hand-off of a packed line batch to `dst` (one xGMI hop on GPU)."""
    hdr = torch.tensor(
        [lines.shape[0], lines.shape[1], 0], dtype=torch.int64, device=lines.device
    )
    dist.send(hdr, dst=dst, tag=tag)
    dist.send(lines.contiguous(), dst=dst, tag=tag + 1)
    dist.send(lens.contiguous(), dst=dst, tag=tag + 2)


def recv_packed(src: int, device: torch.device, tag: int = 0) -> Tuple[torch.Tensor, torch.Tensor]:
    hdr = torch.zeros(_HDR_LEN, dtype=torch.int64, device=device)
    dist.recv(hdr, src=src, tag=tag)
    B, max_len = int(hdr[0]), int(hdr[1])
    lines = torch.empty((B, max_len), dtype=torch.uint8, device=device)
    lens = torch.empty((B,), dtype=torch.int32, device=device)
    dist.recv(lines, src=src, tag=tag + 1)
    dist.recv(lens, src=src, tag=tag + 2)
    return lines, lens


def broadcast_packed(
    lines: Optional[torch.Tensor],
    lens: Optional[torch.Tensor],
    src: int,
    device: torch.device,
    group=None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """1→N fan-out of a packed batch (the reference's multi_output
    broadcast, engine.py:266-302, as ONE collective over xGMI)."""
    rank = dist.get_rank(group) if group is not None else dist.get_rank()
    if rank == src:
        hdr = torch.tensor(
            [lines.shape[0], lines.shape[1], 0], dtype=torch.int64, device=device
        )
    else:
        hdr = torch.zeros(_HDR_LEN, dtype=torch.int64, device=device)
    dist.broadcast(hdr, src=src, group=group)
    B, max_len = int(hdr[0]), int(hdr[1])
    if rank != src:
        lines = torch.empty((B, max_len), dtype=torch.uint8, device=device)
        lens = torch.empty((B,), dtype=torch.int32, device=device)
    dist.broadcast(lines, src=src, group=group)
    dist.broadcast(lens, src=src, group=group)
    return lines, lens


def all_gather_summaries(summary: torch.Tensor, group=None) -> torch.Tensor:
    """All-gather per-rank detector summaries (anomaly counts / score
    moments). Returns [world, len(summary)]."""
    world = dist.get_world_size(group)
    out = [torch.empty_like(summary) for _ in range(world)]
    dist.all_gather(out, summary, group=group)
    return torch.stack(out)


def all_reduce_hashsets(tables: torch.Tensor, group=None) -> None:
    """Merge data-parallel NewValue hash sets across ranks.

    Open-addressing tables cannot be unioned by elementwise max directly
    (slots differ per insertion order), so gather all tables and re-insert
    locally. Intended for low-frequency sync (end of training phase)."""
    world = dist.get_world_size(group)
    gathered = [torch.empty_like(tables) for _ in range(world)]
    dist.all_gather(gathered, tables, group=group)
    from .. import ops

    W, cap = tables.shape
    for other in gathered:
        if other.data_ptr() == tables.data_ptr():
            continue
        # re-insert non-empty keys from `other`
        for w in range(W):
            keys = other[w][other[w] != 0]
            if keys.numel() == 0:
                continue
            h = torch.zeros((keys.numel(), W), dtype=torch.int64, device=tables.device)
            h[:, w] = keys
            if tables.is_cuda:
                from ..ops import _dmx_C

                _dmx_C.hashset_insert(h, tables)
            else:  # pragma: no cover - CPU tables are python sets
                raise RuntimeError("all_reduce_hashsets expects tensor tables")
