"""DistributedPipeline: pipeline stages placed across GPUs/ranks.

Placement is config, not code: a role map assigns each
rank a stage role; packed batches move between stages with the P2P /
broadcast helpers in dist.py (RCCL over xGMI on GPU, gloo on CPU).

Roles:
  "source"    — rank that ingests/creates packed batches (reader edge)
  "parser"    — runs the template-match kernel, forwards match results
  "detector"  — runs NewValue probe + transformer scorer
A rank can hold several roles (collapsed single-process fast path =
pipeline.GpuPipeline). The shipped topologies mirror BASELINE configs 3/4:
  parser rank 0 → broadcast → N detector ranks        (fan-out)
  N independent full pipelines + summary all-gather   (data parallel)
"""
from __future__ import annotations

from typing import Dict, Optional, Sequence, Tuple

import torch
import torch.distributed as tdist

from . import dist as dmx_dist
from ..pipeline import GpuPipeline, PipelineConfig


class DataParallelPipeline:
    """BASELINE config 4: each rank runs the full fused pipeline on its
    shard; per-batch summaries are all-gathered (RCCL over xGMI)."""

    def __init__(self, config: PipelineConfig, device: torch.device) -> None:
        self.pipe = GpuPipeline(config, device=device)
        self.device = device

    def process_packed(self, lines: torch.Tensor, lens: torch.Tensor):
        out = self.pipe.process_packed(lines, lens)
        summary = torch.stack(
            [out["anomaly"].sum().float(), out["scores"].sum().float()]
        ).to(self.device)
        if tdist.is_initialized() and tdist.get_world_size() > 1:
            out["all_summaries"] = dmx_dist.all_gather_summaries(summary)
        else:
            out["all_summaries"] = summary.unsqueeze(0)
        return out


class FanOutPipeline:
    """BASELINE config 3: rank `src` parses, broadcasts the packed batch +
    event ids to every detector rank; detector ranks each run their
    detector stage on the broadcast batch (the reference's multi_output
    broadcast semantics, engine.py:266-302, as one RCCL broadcast)."""

    def __init__(
        self,
        config: PipelineConfig,
        device: torch.device,
        src_rank: int = 0,
    ) -> None:
        self.config = config
        self.device = device
        self.src = src_rank
        self.rank = tdist.get_rank() if tdist.is_initialized() else 0
        # every rank builds the stages it plays
        self.pipe = GpuPipeline(config, device=device)

    def step_source(self, lines: torch.Tensor, lens: torch.Tensor):
        assert self.rank == self.src
        lines, lens = dmx_dist.broadcast_packed(lines, lens, self.src, self.device)
        return self.pipe.process_packed(lines, lens)

    def step_sink(self):
        lines, lens = dmx_dist.broadcast_packed(None, None, self.src, self.device)
        return self.pipe.process_packed(lines, lens)


class StagePipeline:
    """Two-stage P2P placement: parser rank forwards the packed batch to a
    detector rank (reader→parser→detector across GPUs, one xGMI hop per
    edge — BASELINE config 2 split across devices)."""

    def __init__(self, config: PipelineConfig, device: torch.device,
                 parser_rank: int = 0, detector_rank: int = 1) -> None:
        self.config = config
        self.device = device
        self.parser_rank = parser_rank
        self.detector_rank = detector_rank
        self.rank = tdist.get_rank() if tdist.is_initialized() else 0
        self.pipe = GpuPipeline(config, device=device)

    def step_parser(self, lines: torch.Tensor, lens: torch.Tensor):
        assert self.rank == self.parser_rank
        match = self.pipe.matcher.match_packed(lines, lens)
        dmx_dist.send_packed(lines, lens, dst=self.detector_rank)
        ev = match["event_id"]
        tdist.send(ev.to(torch.int32).contiguous(), dst=self.detector_rank, tag=9)
        return match

    def step_detector(self):
        assert self.rank == self.detector_rank
        lines, lens = dmx_dist.recv_packed(self.parser_rank, self.device)
        ev = torch.empty((lines.shape[0],), dtype=torch.int32, device=self.device)
        tdist.recv(ev, src=self.parser_rank, tag=9)
        # run the full fused pipeline on the received batch (detector stage
        # includes its own match for capture spans; event ids crosschecked)
        out = self.pipe.process_packed(lines, lens)
        out["forwarded_event_id"] = ev
        return out
