"""TransformerDetector: BERT-tiny bf16 MFMA log-anomaly detector component.

The service-level face of BASELINE.json config 5: a ``CoreDetector`` whose
``process_batch`` decodes ParserSchema frames, packs the raw log lines
into the GPU SoA buffer, scores them with the byte-level BERT-tiny model
(hand-written MFMA kernels via ``detectmateservice_amd.ops``) and emits a
DetectorSchema alert for every line whose score exceeds the calibrated
threshold. During the ``data_use_training`` phase it collects score
statistics and sets ``threshold = mean + z_threshold * std``.

This is a capability the reference library family carries via its ML deps
(tiktoken/openai/sklearn in detectmatelibrary's dependency set, SURVEY.md
§2.2 "capability hints") rebuilt MI355X-first: batched, GPU-resident,
checkpointable (state_dict covers weights + calibration).
"""
from __future__ import annotations

import math
import time
from typing import Any, Dict, List, Optional

import torch

from ...components.base import CoreDetector, CoreDetectorConfig
from ...models.bert_tiny import BertTinyConfig, BertTinyDetectorModel
from ...schemas import DetectorSchema, ParserSchema
from ... import ops


class TransformerDetectorConfig(CoreDetectorConfig):
    method_type: str = "transformer_detector"
    auto_config: bool = False
    params: Dict[str, Any] = {}
    #: z-score threshold over the training-score distribution
    z_threshold: float = 3.0
    max_seq: int = 64
    hidden: int = 128
    layers: int = 2
    heads: int = 2
    ffn: int = 512
    device: Optional[str] = None
    seed: int = 1234
    batch_max_len: int = 256


class TransformerDetector(CoreDetector):
    CONFIG_CLASS = TransformerDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        cfg = self.config
        if cfg.device:
            self.device = torch.device(cfg.device)
        else:
            self.device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
        self.model = BertTinyDetectorModel(
            BertTinyConfig(
                hidden=cfg.hidden,
                layers=cfg.layers,
                heads=cfg.heads,
                ffn=cfg.ffn,
                max_seq=cfg.max_seq,
            ),
            device=self.device,
            seed=cfg.seed,
        )
        self.detector_id = f"transformer_detector-{id(self):x}"
        # calibration state
        self._score_sum = 0.0
        self._score_sq = 0.0
        self._score_n = 0
        self.threshold: Optional[float] = None

    # ------------------------------------------------------------------
    def _score_lines(self, raw_lines: List[bytes]) -> torch.Tensor:
        lines, lens = ops.pack_lines(
            raw_lines, self.config.batch_max_len, device=self.device
        )
        start = torch.zeros(len(raw_lines), dtype=torch.int32, device=self.device)
        # score_spans takes the fused whole-model kernel on GPU at the
        # flagship geometry, layered kernels / CPU reference otherwise
        return self.model.score_spans(lines, start, lens.int())

    def train(self, parsed_batch: List[ParserSchema]) -> None:
        raws = [(p.log or "").encode() for p in parsed_batch]
        scores = self._score_lines(raws).float().cpu()
        self._score_sum += float(scores.sum())
        self._score_sq += float((scores ** 2).sum())
        self._score_n += len(raws)
        if self._score_n > 1:
            mean = self._score_sum / self._score_n
            var = max(self._score_sq / self._score_n - mean * mean, 1e-12)
            self.threshold = mean + self.config.z_threshold * math.sqrt(var)

    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]:
        parsed = [ParserSchema.deserialize(f) for f in frames]
        n_train = int(getattr(self.config, "data_use_training", 0))
        results: List[Optional[bytes]] = [None] * len(frames)

        train_upto = 0
        if self._seen_lines < n_train:
            train_upto = min(len(parsed), n_train - self._seen_lines)
            self.train(parsed[:train_upto])
        self._seen_lines += len(parsed)
        rest = parsed[train_upto:]
        if not rest:
            return results

        scores = self._score_lines([(p.log or "").encode() for p in rest])
        thr = self.threshold if self.threshold is not None else float("inf")
        flags = (scores > thr).cpu().tolist()
        scores_cpu = scores.float().cpu().tolist()
        now = int(time.time())
        for j, (p, hit) in enumerate(zip(rest, flags)):
            if not hit:
                continue
            alert = DetectorSchema(
                detectorID=self.detector_id,
                detectorType="transformer_detector",
                alertID=f"tf-{p.logID or p.parsedLogID}",
                detectionTimestamp=now,
                logIDs=[p.logID] if p.logID else [],
                score=float(scores_cpu[j]),
                description=f"Anomalous line (score {scores_cpu[j]:.3f} > {thr:.3f})",
            )
            results[train_upto + j] = alert.serialize()
        return results

    def train_noop(self):  # CoreDetector abstract satisfied via train above
        pass

    def detect(self, parsed, alert) -> bool:  # per-frame API fallback
        out = self.process_batch([parsed.serialize()])
        if out[0] is None:
            return False
        got = DetectorSchema.deserialize(out[0])
        for f in alert.FIELDS:
            setattr(alert, f, getattr(got, f))
        return True

    # ------------------------------------------------------------------
    def state_dict(self) -> Dict[str, Any]:
        return {
            "seen_lines": self._seen_lines,
            "score_sum": self._score_sum,
            "score_sq": self._score_sq,
            "score_n": self._score_n,
            "threshold": self.threshold,
            "model": self.model.state_dict(),
        }

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self._seen_lines = int(state.get("seen_lines", 0))
        self._score_sum = float(state.get("score_sum", 0.0))
        self._score_sq = float(state.get("score_sq", 0.0))
        self._score_n = int(state.get("score_n", 0))
        self.threshold = state.get("threshold")
        if "model" in state:
            self.model.load_state_dict(state["model"])
