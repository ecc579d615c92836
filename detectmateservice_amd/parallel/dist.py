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


def init_from_env(backend: Optional[str] = None,
                  timeout_s: float = 120.0) -> Tuple[int, int]:
    """Initialize the default process group from torchrun env vars.

    Returns (rank, world_size); no-op (0, 1) when WORLD_SIZE is unset.
    Backend defaults to nccl (=RCCL) when a GPU is visible, else gloo.
    The timeout bounds collective waits so a dead peer surfaces as an
    error the engine loops can handle (drop-don't-block posture)."""
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return 0, 1
    rank = int(os.environ.get("RANK", "0"))
    if not dist.is_initialized():
        import datetime

        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world_size,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
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
    (slots differ per insertion order), so gather all tables and
    re-insert every peer table's keys with ONE batched insert-kernel
    call per peer: the transposed table [capacity, W] IS a valid hash
    matrix for dmx_hashset_insert (column w holds watch w's keys; empty
    slots are 0, which the kernel skips). Intended for low-frequency
    sync (end of training phase / explicit dp_sync)."""
    world = dist.get_world_size(group)
    if world <= 1:
        return
    if not tables.is_cuda:  # pragma: no cover - CPU tables are python sets
        raise RuntimeError("all_reduce_hashsets expects device tensor tables")
    # gloo cannot all_gather CUDA tensors (mixed gloo + GPU deployments:
    # several services sharing one GPU): stage through host copies there
    backend = dist.get_backend(group)
    comm = tables if backend == "nccl" else tables.cpu()
    gathered = [torch.empty_like(comm) for _ in range(world)]
    dist.all_gather(gathered, comm, group=group)
    from ..ops import _dmx_C

    me = dist.get_rank(group)
    for r, other in enumerate(gathered):
        if r == me:
            continue
        _dmx_C.hashset_insert(
            other.t().contiguous().to(tables.device), tables)


# ---------------------------------------------------------------------------
# frame transport for settings-driven service placement (dist_mode):
# serialized protobuf frames as one u8 blob + offsets, moved by
# broadcast (fanout) or P2P (stage). A 3-word header precedes each
# batch; flag 1 = heartbeat (no frames, keeps sinks responsive to
# stop), flag 2 = shutdown sentinel.
# ---------------------------------------------------------------------------

FRAME_DATA = 0
FRAME_HEARTBEAT = 1
FRAME_SHUTDOWN = 2


def _frames_to_tensors(frames: Sequence[bytes]):
    offs = torch.zeros(len(frames) + 1, dtype=torch.int64)
    total = 0
    for i, f in enumerate(frames):
        total += len(f)
        offs[i + 1] = total
    blob = b"".join(frames)
    data = (
        torch.frombuffer(bytearray(blob), dtype=torch.uint8)
        if blob
        else torch.zeros(0, dtype=torch.uint8)
    )
    return data, offs


def _tensors_to_frames(data: torch.Tensor, offs: torch.Tensor) -> List[bytes]:
    raw = bytes(data.cpu().numpy().tobytes())
    o = offs.tolist()
    return [raw[o[i]:o[i + 1]] for i in range(len(o) - 1)]


def broadcast_frames_src(frames: Sequence[bytes], src: int,
                         device: torch.device, flag: int = FRAME_DATA,
                         group=None) -> None:
    """Source side of the fanout hop (reference multi_output broadcast,
    engine.py:266-302, as ONE collective over xGMI/gloo)."""
    data, offs = _frames_to_tensors(frames)
    hdr = torch.tensor([len(frames), data.numel(), flag], dtype=torch.int64,
                       device=device)
    dist.broadcast(hdr, src=src, group=group)
    if flag == FRAME_DATA and len(frames) > 0:
        dist.broadcast(offs.to(device), src=src, group=group)
        dist.broadcast(data.to(device), src=src, group=group)


def broadcast_frames_sink(src: int, device: torch.device,
                          group=None) -> Tuple[List[bytes], int]:
    hdr = torch.zeros(3, dtype=torch.int64, device=device)
    dist.broadcast(hdr, src=src, group=group)
    n, nbytes, flag = int(hdr[0]), int(hdr[1]), int(hdr[2])
    if flag != FRAME_DATA or n == 0:
        return [], flag
    offs = torch.zeros(n + 1, dtype=torch.int64, device=device)
    dist.broadcast(offs, src=src, group=group)
    data = torch.zeros(nbytes, dtype=torch.uint8, device=device)
    dist.broadcast(data, src=src, group=group)
    return _tensors_to_frames(data, offs), flag


def send_frames(frames: Sequence[bytes], dst: int, device: torch.device,
                flag: int = FRAME_DATA, tag: int = 100) -> None:
    """P2P stage hop (one xGMI link on GPU — SURVEY.md §5.8)."""
    data, offs = _frames_to_tensors(frames)
    hdr = torch.tensor([len(frames), data.numel(), flag], dtype=torch.int64,
                       device=device)
    dist.send(hdr, dst=dst, tag=tag)
    if flag == FRAME_DATA and len(frames) > 0:
        dist.send(offs.to(device), dst=dst, tag=tag + 1)
        dist.send(data.to(device), dst=dst, tag=tag + 2)


def recv_frames(src: int, device: torch.device,
                tag: int = 100) -> Tuple[List[bytes], int]:
    hdr = torch.zeros(3, dtype=torch.int64, device=device)
    dist.recv(hdr, src=src, tag=tag)
    n, nbytes, flag = int(hdr[0]), int(hdr[1]), int(hdr[2])
    if flag != FRAME_DATA or n == 0:
        return [], flag
    offs = torch.zeros(n + 1, dtype=torch.int64, device=device)
    dist.recv(offs, src=src, tag=tag + 1)
    data = torch.zeros(nbytes, dtype=torch.uint8, device=device)
    dist.recv(data, src=src, tag=tag + 2)
    return _tensors_to_frames(data, offs), flag
