"""GpuPipeline: fused reader→parser→detector on one device.

The MI355X-native fast path: when pipeline stages are
co-located, frames never round-trip through protobuf between stages —
lines live in a [B, max_len] u8 SoA tensor on the GPU and flow
parser-kernel → hash-probe / transformer-scoring as device tensors on one
HIP stream. The socket Service path (core.py) remains for distributed /
edge deployment; this class is what bench.py drives and what a
TransformerDetector service uses internally per batch.

Stage structure per batch (all on `device`):
  1. template_match kernel: log_format header split + template match
     (event_id + capture spans)
  2. NewValue watch: hash watched spans (kernel) -> probe GPU hash sets
     (training batches insert instead)
  3. Transformer scorer (BERT-tiny bf16 MFMA) over content-span byte tokens
Anomaly = NewValue unseen-value hit OR transformer score > threshold.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

import torch

from . import ops
from .models.bert_tiny import BertTinyConfig, BertTinyDetectorModel


@dataclass
class PipelineConfig:
    templates: Sequence[str] = ()
    log_format: Optional[str] = None
    lowercase: bool = False
    max_len: int = 512
    #: watched fields: list of dicts {kind: "variable"|"header", pos: int,
    #: event: int (-1 = any)} — mirrors NewValueDetector specs
    watches: Sequence[dict] = ()
    hashset_capacity: int = 1 << 16
    #: transformer detector on/off + threshold
    use_transformer: bool = True
    score_threshold: float = 3.0
    train_lines: int = 0  # NewValue training-phase length (data_use_training)
    bert: BertTinyConfig = field(default_factory=BertTinyConfig)
    seed: int = 1234


class GpuPipeline:
    def __init__(self, config: PipelineConfig, device: str | torch.device = "cpu") -> None:
        self.config = config
        self.device = torch.device(device)
        self.matcher = ops.TemplateMatcher(
            config.templates,
            log_format=config.log_format,
            lowercase=config.lowercase,
            device=self.device,
            max_len=config.max_len,
        )
        specs = []
        for w in config.watches:
            kind = 0 if w.get("kind", "variable") == "variable" else 1
            specs.append([kind, int(w.get("event", -1)), int(w["pos"]), 0])
        self.specs = (
            torch.tensor(specs, dtype=torch.int32, device=self.device)
            if specs
            else torch.zeros((0, 4), dtype=torch.int32, device=self.device)
        )
        self.hashsets = (
            ops.GpuHashSets(len(specs), config.hashset_capacity, device=self.device)
            if specs
            else None
        )
        self.model = (
            BertTinyDetectorModel(config.bert, device=self.device, seed=config.seed)
            if config.use_transformer
            else None
        )
        self.seen_lines = 0
        # hipGraph capture state (enable_graph(); steady-state detect only)
        self._graph = None
        self._graph_in: Optional[tuple] = None
        self._graph_out: Optional[Dict[str, torch.Tensor]] = None

    # ------------------------------------------------------------------
    def process_packed(
        self, lines: torch.Tensor, line_len: torch.Tensor
    ) -> Dict[str, torch.Tensor]:
        """lines [B, max_len] u8 + lengths on self.device → result tensors.

        Returns event_id [B], anomaly [B] (bool), scores [B] (f32, 0 when
        transformer off), nv_unseen [B, W]."""
        B = lines.shape[0]
        if B == 0:
            return {
                "event_id": torch.zeros(0, dtype=torch.int32, device=lines.device),
                "anomaly": torch.zeros(0, dtype=torch.bool, device=lines.device),
                "scores": torch.zeros(0, dtype=torch.float32, device=lines.device),
                "nv_unseen": None,
                "match": None,
            }
        # roctx-visible stage ranges (torch.cuda.nvtx maps to rocTracer
        # markers on ROCm — SURVEY.md §5.1 tracing requirement)
        _rng = torch.cuda.nvtx.range if lines.is_cuda else None
        if _rng:
            torch.cuda.nvtx.range_push("dmx::parse")
        match = self.matcher.match_packed(lines, line_len)
        if _rng:
            torch.cuda.nvtx.range_pop()

        n_train_left = max(0, self.config.train_lines - self.seen_lines)
        train_upto = min(B, n_train_left)
        self.seen_lines += B

        nv_unseen = None
        if _rng:
            torch.cuda.nvtx.range_push("dmx::new_value")
        if self.hashsets is not None and self.specs.shape[0] > 0:
            hashes = ops.watch_hashes(lines, match, self.specs, self.config.lowercase)
            if train_upto > 0:
                self.hashsets.insert(hashes[:train_upto])
            if train_upto == 0:
                # steady state: probe output IS the result (a zeros +
                # full-copy here cost 2 kernel launches per batch)
                nv_unseen = self.hashsets.probe(hashes)
            elif train_upto < B:
                unseen_tail = self.hashsets.probe(hashes[train_upto:])
                nv_unseen = torch.zeros(
                    (B, self.specs.shape[0]), dtype=torch.int32, device=lines.device
                )
                nv_unseen[train_upto:] = unseen_tail
            else:
                nv_unseen = torch.zeros(
                    (B, self.specs.shape[0]), dtype=torch.int32, device=lines.device
                )

        if _rng:
            torch.cuda.nvtx.range_pop()

        scores = torch.zeros(B, dtype=torch.float32, device=lines.device)
        if _rng:
            torch.cuda.nvtx.range_push("dmx::transformer")
        if self.model is not None:
            # content span comes straight from the match kernel
            # (span_start/span_end outputs — the former gather/where
            # chain here was ~6 kernel launches per batch, profiles/r10)
            scores = self.model.score_spans(
                lines, match["span_start"], match["span_end"])
        if _rng:
            torch.cuda.nvtx.range_pop()

        anomaly = scores > self.config.score_threshold
        if nv_unseen is not None:
            anomaly = anomaly | (nv_unseen.sum(dim=1) > 0)
        return {
            "event_id": match["event_id"],
            "anomaly": anomaly,
            "scores": scores,
            "nv_unseen": nv_unseen,
            "match": match,
        }

    def process_lines(self, raw_lines: Sequence[bytes]) -> Dict[str, torch.Tensor]:
        lines, lens = ops.pack_lines(raw_lines, self.config.max_len, device=self.device)
        return self.process_packed(lines, lens)

    # ------------------------------------------------------------------
    # hipGraph capture (HIP graphs for the launch-bound steady state —
    # SURVEY.md build mandate). Valid only AFTER the training phase: the
    # captured graph replays the detect-only path at fixed batch size.
    # ------------------------------------------------------------------
    def enable_graph(self, batch_size: int) -> bool:
        if self.device.type != "cuda":
            return False
        if self.seen_lines < self.config.train_lines:
            raise RuntimeError("enable_graph() requires the training phase done")
        B = batch_size
        static_lines = torch.zeros(
            (B, self.config.max_len), dtype=torch.uint8, device=self.device
        )
        static_lens = torch.zeros((B,), dtype=torch.int32, device=self.device)
        # warm up allocator/kernels on the same shapes, then capture
        for _ in range(2):
            self.process_packed(static_lines, static_lens)
            self.seen_lines -= B  # warmups must not advance stream state
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            out = self.process_packed(static_lines, static_lens)
        self.seen_lines -= B
        self._graph = graph
        self._graph_in = (static_lines, static_lens)
        self._graph_out = out
        return True

    def process_packed_graph(
        self, lines: torch.Tensor, lens: torch.Tensor
    ) -> Dict[str, torch.Tensor]:
        """Replay the captured graph on a new batch (same B, max_len)."""
        assert self._graph is not None, "call enable_graph() first"
        sl, sn = self._graph_in
        sl.copy_(lines, non_blocking=True)
        sn.copy_(lens, non_blocking=True)
        self._graph.replay()
        self.seen_lines += lines.shape[0]
        return self._graph_out

    def process_packed_graph_partial(
        self, lines: torch.Tensor, lens: torch.Tensor
    ) -> Dict[str, torch.Tensor]:
        """Replay on a batch SMALLER than the captured size: pad rows get
        len 0 (their kernels no-op on the span but the transformer still
        runs them — acceptable when partial batches are rare stream
        edges). Outputs are sliced back to the true batch size."""
        assert self._graph is not None, "call enable_graph() first"
        sl, sn = self._graph_in
        B = lines.shape[0]
        assert B <= sl.shape[0]
        sl[:B].copy_(lines, non_blocking=True)
        sn[:B].copy_(lens, non_blocking=True)
        if B < sn.shape[0]:
            sn[B:].zero_()
        self._graph.replay()
        self.seen_lines += B

        def cut(v):
            if torch.is_tensor(v):
                return v[:B]
            if isinstance(v, dict):
                return {k: cut(x) for k, x in v.items()}
            return v

        return {k: cut(v) for k, v in self._graph_out.items()}
