"""EmbeddingDetector: transformer-embedding distance anomaly detection.

The reference library family carries embedding/LLM-assisted detection
(its dependency set includes tiktoken/openai — SURVEY.md §2.2 "capability
hints"). MI355X-native rebuild: log lines are embedded with the BERT-tiny
byte transformer (mean-pooled hidden state, bf16 MFMA kernels); training
fits a diagonal Gaussian over embeddings; detection flags lines whose
normalized distance

    d(x) = sqrt(mean_i ((x_i - mu_i)^2 / (var_i + eps)))

exceeds ``z_threshold``. Fully checkpointable (weights + moments).
"""
from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

import torch

from ...components.base import CoreDetector, CoreDetectorConfig
from ...models.bert_tiny import BertTinyConfig, BertTinyDetectorModel
from ...schemas import DetectorSchema, ParserSchema
from ... import ops


class EmbeddingDetectorConfig(CoreDetectorConfig):
    method_type: str = "embedding_detector"
    z_threshold: float = 4.0
    max_seq: int = 64
    hidden: int = 128
    layers: int = 2
    heads: int = 2
    ffn: int = 512
    device: Optional[str] = None
    seed: int = 1234
    batch_max_len: int = 256


class EmbeddingDetector(CoreDetector):
    CONFIG_CLASS = EmbeddingDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        cfg = self.config
        self.device = torch.device(
            cfg.device or ("cuda" if torch.cuda.is_available() else "cpu")
        )
        self.model = BertTinyDetectorModel(
            BertTinyConfig(hidden=cfg.hidden, layers=cfg.layers,
                           heads=cfg.heads, ffn=cfg.ffn, max_seq=cfg.max_seq),
            device=self.device, seed=cfg.seed,
        )
        self.detector_id = f"embedding_detector-{id(self):x}"
        h = cfg.hidden
        self._sum = torch.zeros(h, dtype=torch.float64)
        self._sumsq = torch.zeros(h, dtype=torch.float64)
        self._n = 0

    # ------------------------------------------------------------------
    def embed(self, raw_lines: List[bytes]) -> torch.Tensor:
        """Mean-pooled transformer embeddings [B, hidden] (f32)."""
        lines, lens = ops.pack_lines(
            raw_lines, self.config.batch_max_len, device=self.device
        )
        start = torch.zeros(len(raw_lines), dtype=torch.int32, device=self.device)
        tokens = self.model.tokenize_spans(lines, start, lens.int())
        c = self.model.config
        x = self.model.tok_emb[tokens.long()] + self.model.pos_emb[: c.max_seq].unsqueeze(0)
        # full forward reusing the layered/fused ops, capturing pooled state
        scores_unused = None
        # run the transformer body (same math as forward, minus score head)
        x = x.to(self.model.dtype).contiguous()
        B, S = tokens.shape
        M = B * S
        h = c.hidden
        for layer in self.model.layers:
            x2 = x.view(M, h)
            qkv = ops.fused_linear(x2, layer["wqkv_t"], layer["bqkv"])
            attn = ops.attention_qkv(qkv.view(B, S, 3 * h), S, c.heads, c.head_dim).view(M, h)
            proj = ops.fused_linear(attn, layer["wo_t"], layer["bo"])
            x1 = ops.layernorm(proj, layer["ln1_g"], layer["ln1_b"], residual=x2)
            ffn = ops.fused_linear(x1, layer["w1_t"], layer["b1"], activation="gelu")
            ffn = ops.fused_linear(ffn, layer["w2_t"], layer["b2"])
            x = ops.layernorm(ffn, layer["ln2_g"], layer["ln2_b"], residual=x1).view(B, S, h)
        return x.float().mean(dim=1).cpu()

    def train(self, parsed_batch: List[ParserSchema]) -> None:
        emb = self.embed([(p.log or "").encode() for p in parsed_batch])
        self._sum += emb.double().sum(dim=0)
        self._sumsq += (emb.double() ** 2).sum(dim=0)
        self._n += emb.shape[0]

    def _distance(self, emb: torch.Tensor) -> torch.Tensor:
        n = max(self._n, 1)
        mu = (self._sum / n).float()
        var = (self._sumsq / n).float() - mu ** 2
        z2 = (emb - mu) ** 2 / (var.clamp(min=1e-6))
        return z2.mean(dim=1).sqrt()

    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]:
        parsed = [ParserSchema.deserialize(f) for f in frames]
        results: List[Optional[bytes]] = [None] * len(frames)
        n_train = int(self.config.data_use_training)
        train_upto = 0
        if self._seen_lines < n_train:
            train_upto = min(len(parsed), n_train - self._seen_lines)
            self.train(parsed[:train_upto])
        self._seen_lines += len(parsed)
        rest = parsed[train_upto:]
        if not rest or self._n == 0:
            return results
        emb = self.embed([(p.log or "").encode() for p in rest])
        dist = self._distance(emb)
        now = int(time.time())
        thr = float(self.config.z_threshold)
        for j, (p, d) in enumerate(zip(rest, dist.tolist())):
            if d <= thr:
                continue
            results[train_upto + j] = DetectorSchema(
                detectorID=self.detector_id,
                detectorType="embedding_detector",
                alertID=f"emb-{p.logID or p.parsedLogID}",
                detectionTimestamp=now,
                logIDs=[p.logID] if p.logID else [],
                score=float(d),
                description=f"Embedding distance {d:.3f} > {thr:.3f}",
            ).serialize()
        return results

    def detect(self, parsed, alert) -> bool:
        out = self.process_batch([parsed.serialize()])
        if out[0] is None:
            return False
        got = DetectorSchema.deserialize(out[0])
        for f in alert.FIELDS:
            setattr(alert, f, getattr(got, f))
        return True

    def state_dict(self) -> Dict[str, Any]:
        return {
            "seen_lines": self._seen_lines,
            "sum": self._sum, "sumsq": self._sumsq, "n": self._n,
            "model": self.model.state_dict(),
        }

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self._seen_lines = int(state.get("seen_lines", 0))
        self._sum = state["sum"]
        self._sumsq = state["sumsq"]
        self._n = int(state["n"])
        self.model.load_state_dict(state["model"])
