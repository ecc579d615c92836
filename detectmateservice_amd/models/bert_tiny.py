"""ByteBERT-tiny log-anomaly scorer (BASELINE.json config 5).

A BERT-tiny-shaped transformer (2 layers, hidden 128, 2 heads, FFN 512)
over byte-level tokens of the parsed log content, scoring each line's
anomaly likelihood. All dense compute runs through the hand-written CDNA4
kernels in ``detectmateservice_amd.ops`` (MFMA fused linear, fused
residual+LayerNorm, fused short-seq attention); embedding gather and the
final scalar head use plain torch (PyTorch-ROCm is the sanctioned compute
path for non-hot ops).

Weights are stored PRE-TRANSPOSED ([out, in] = [N, K]) because the MFMA
kernel wants both operands K-contiguous (ops/csrc/gemm_bf16.hip header).
Random-init weights (no network for checkpoints — BASELINE.md).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch

from .. import ops


@dataclass
class BertTinyConfig:
    vocab_size: int = 259  # 256 byte values + pad/cls/unused
    hidden: int = 128
    layers: int = 2
    heads: int = 2
    ffn: int = 512
    max_seq: int = 64
    pad_id: int = 0

    @property
    def head_dim(self) -> int:
        return self.hidden // self.heads


class BertTinyDetectorModel:
    """Inference-first model over custom ops (no autograd graph needed for
    the streaming detector; training/calibration uses score statistics)."""

    def __init__(
        self,
        config: BertTinyConfig = BertTinyConfig(),
        device: str | torch.device = "cpu",
        dtype: torch.dtype = torch.bfloat16,
        seed: int = 1234,
    ) -> None:
        self.config = config
        self.device = torch.device(device)
        self.dtype = dtype
        g = torch.Generator().manual_seed(seed)
        c = config
        h, f = c.hidden, c.ffn

        def rnd(*shape, std=0.02):
            return (torch.randn(*shape, generator=g) * std).to(dtype)

        self.tok_emb = rnd(c.vocab_size, h).to(self.device)
        self.pos_emb = rnd(c.max_seq, h).to(self.device)
        self.layers = []
        for _ in range(c.layers):
            layer = {
                "wqkv_t": rnd(3 * h, h).to(self.device),   # [N=3h, K=h]
                "bqkv": torch.zeros(3 * h, dtype=torch.float32, device=self.device),
                "wo_t": rnd(h, h).to(self.device),
                "bo": torch.zeros(h, dtype=torch.float32, device=self.device),
                "ln1_g": torch.ones(h, dtype=dtype, device=self.device),
                "ln1_b": torch.zeros(h, dtype=dtype, device=self.device),
                "w1_t": rnd(f, h).to(self.device),
                "b1": torch.zeros(f, dtype=torch.float32, device=self.device),
                "w2_t": rnd(h, f).to(self.device),
                "b2": torch.zeros(h, dtype=torch.float32, device=self.device),
                "ln2_g": torch.ones(h, dtype=dtype, device=self.device),
                "ln2_b": torch.zeros(h, dtype=dtype, device=self.device),
            }
            self.layers.append(layer)
        self.w_score = rnd(h, 1, std=0.1).to(self.device)  # [h, 1]
        self.b_score = torch.zeros(1, dtype=torch.float32, device=self.device)

    # ------------------------------------------------------------------
    def forward(self, tokens: torch.Tensor) -> torch.Tensor:
        """tokens [B, S] int64/int32 -> anomaly scores [B] (f32)."""
        c = self.config
        B, S = tokens.shape
        assert S <= c.max_seq
        x = self.tok_emb[tokens.long()] + self.pos_emb[:S].unsqueeze(0)
        x = x.to(self.dtype).contiguous()  # [B, S, h]
        M = B * S
        h = c.hidden

        for layer in self.layers:
            x2 = x.view(M, h)
            qkv = ops.fused_linear(x2, layer["wqkv_t"], layer["bqkv"])  # [M, 3h]
            # Fused-QKV attention: the GPU kernel reads [B, S, 3*H*Dh]
            # directly (no permutes). NOTE the qkv linear emits q|k|v
            # CONCATENATED over N: row = [q(h0..hH), k(...), v(...)] which
            # is exactly the 3*H*Dh layout the kernel indexes.
            attn = ops.attention_qkv(
                qkv.view(B, S, 3 * h), S, c.heads, c.head_dim
            ).view(M, h)
            proj = ops.fused_linear(attn, layer["wo_t"], layer["bo"])
            x1 = ops.layernorm(proj, layer["ln1_g"], layer["ln1_b"], residual=x2)
            ffn = ops.fused_linear(x1, layer["w1_t"], layer["b1"], activation="gelu")
            ffn = ops.fused_linear(ffn, layer["w2_t"], layer["b2"])
            x = ops.layernorm(ffn, layer["ln2_g"], layer["ln2_b"], residual=x1)
            x = x.view(B, S, h)

        pooled = x.float().mean(dim=1)  # [B, h]
        scores = pooled @ self.w_score.float() + self.b_score
        return scores.squeeze(-1)

    __call__ = forward

    # ------------------------------------------------------------------
    def tokenize_spans(
        self,
        lines: torch.Tensor,   # [B, max_len] u8 (device)
        start: torch.Tensor,   # [B] i32 content span start
        end: torch.Tensor,     # [B] i32 content span end
    ) -> torch.Tensor:
        """Byte-level tokens of each line's content span: id = byte + 3,
        pad 0. Pure tensor ops (runs on GPU, no host roundtrip)."""
        c = self.config
        B = lines.shape[0]
        S = c.max_seq
        idx = start.long().unsqueeze(1) + torch.arange(S, device=lines.device).unsqueeze(0)
        valid = idx < end.long().unsqueeze(1)
        idx = idx.clamp(max=lines.shape[1] - 1)
        toks = lines.long().gather(1, idx) + 3
        toks = torch.where(valid, toks, torch.zeros_like(toks))
        return toks

    # ------------------------------------------------------------------
    # fused whole-model kernel path (ops/csrc/bert_fused.hip)
    # ------------------------------------------------------------------
    def _fused_ok(self) -> bool:
        c = self.config
        return (
            self.device.type == "cuda"
            and ops.have_extension()
            and c.hidden == 128
            and c.heads == 2
            and c.ffn == 512
            and c.max_seq == 64
            and c.vocab_size == 259
        )

    def _fused_blobs(self):
        """Pack weights into the (bf16, f32) blobs the fused kernel indexes
        (layout MUST match bert_fused.hip WB_*/FB_* offsets)."""
        if getattr(self, "_wb", None) is None:
            parts = [self.tok_emb, self.pos_emb]
            for layer in self.layers:
                parts += [
                    layer["wqkv_t"], layer["wo_t"], layer["w1_t"], layer["w2_t"],
                    layer["ln1_g"], layer["ln1_b"], layer["ln2_g"], layer["ln2_b"],
                ]
            parts.append(self.w_score.reshape(-1))
            self._wb = torch.cat([p.reshape(-1) for p in parts]).contiguous()
            fparts = []
            for layer in self.layers:
                fparts += [layer["bqkv"], layer["bo"], layer["b1"], layer["b2"]]
            fparts.append(self.b_score)
            self._fb = torch.cat([p.reshape(-1) for p in fparts]).contiguous()
        return self._wb, self._fb

    def score_spans(
        self, lines: torch.Tensor, start: torch.Tensor, end: torch.Tensor
    ) -> torch.Tensor:
        """lines [B, max_len] u8 + content spans -> anomaly scores [B] f32.

        GPU + flagship geometry: ONE fused kernel (embed → N transformer
        layers → pool → score, all in LDS). Otherwise: tokenize + layered
        forward (same numerics at bf16 tolerance — tests/test_gpu_ops.py).
        """
        if lines.shape[0] == 0:
            return torch.zeros(0, dtype=torch.float32, device=lines.device)
        if self._fused_ok():
            from ..ops import _dmx_C  # type: ignore[attr-defined]

            wb, fb = self._fused_blobs()
            return _dmx_C.bert_fused_bf16(
                lines, start.int().contiguous(), end.int().contiguous(),
                wb, fb, self.config.layers, 1e-5,
            )
        tokens = self.tokenize_spans(lines, start.int(), end.int())
        return self.forward(tokens)

    def state_dict(self) -> Dict[str, torch.Tensor]:
        out = {"tok_emb": self.tok_emb, "pos_emb": self.pos_emb,
               "w_score": self.w_score, "b_score": self.b_score}
        for i, layer in enumerate(self.layers):
            for k, v in layer.items():
                out[f"layer{i}.{k}"] = v
        return {k: v.cpu() for k, v in out.items()}

    def load_state_dict(self, state: Dict[str, torch.Tensor]) -> None:
        self._wb = self._fb = None  # invalidate fused blobs
        self.tok_emb = state["tok_emb"].to(self.device)
        self.pos_emb = state["pos_emb"].to(self.device)
        self.w_score = state["w_score"].to(self.device)
        self.b_score = state["b_score"].to(self.device)
        for i, layer in enumerate(self.layers):
            for k in list(layer):
                layer[k] = state[f"layer{i}.{k}"].to(self.device)
