"""Tokenizer- and LLM-assisted detector family.

The reference's library dependency set includes ``tiktoken`` and
``openai`` (SURVEY.md §2.2 capability hints, uv.lock:277-294) — i.e. a
tokenization-based and an LLM-assisted detection family exist in that
ecosystem. This module provides the MI355X framework's equivalents,
designed for this offline environment:

* :class:`TokenizerDetector` — trains a byte-level BPE tokenizer
  (HuggingFace ``tokenizers``) on the training stream; a line's anomaly
  score is its *compression surprise*: tokens-per-byte under the learned
  vocabulary, z-scored against the training distribution. Log lines that
  fit the learned structure tokenize into few long merges; novel
  structure fragments into many short tokens. This is the classic
  compression-based novelty signal the tiktoken dependency hints at,
  with no model weights and no network.

* :class:`LLMAssistDetector` — wraps ANY inner detector and enriches its
  alerts through an OpenAI-compatible ``/v1/chat/completions`` endpoint
  (a local vLLM/llama.cpp server, or nothing: on any error the alert
  passes through unenriched — the LLM is assistance, never the detection
  path). The reference family's ``openai`` dependency maps to exactly
  this shape. Tested offline against a stub HTTP server.
"""
from __future__ import annotations

import json
import math
import time
import urllib.request
from typing import Any, Dict, List, Optional

from ...components.base import CoreDetector, CoreDetectorConfig
from ...schemas import DetectorSchema, ParserSchema


class TokenizerDetectorConfig(CoreDetectorConfig):
    method_type: str = "tokenizer_detector"
    #: BPE vocabulary size learned from the training stream
    vocab_size: int = 2048
    #: alert when the line's tokens-per-byte z-score exceeds this
    z_threshold: float = 4.0
    #: last fraction of the training lines used to calibrate the
    #: tokens-per-byte distribution (the rest trains the BPE merges)
    calibration_fraction: float = 0.2
    params: Dict[str, Any] = {}


class TokenizerDetector(CoreDetector):
    CONFIG_CLASS = TokenizerDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        self._train_lines: List[str] = []
        self._tok = None
        self._mean = 0.0
        self._std = 1.0
        self.detector_id = f"tokenizer_detector-{id(self):x}"

    # ------------------------------------------------------------------
    def _fit(self) -> None:
        from tokenizers import Tokenizer, models, pre_tokenizers, trainers

        lines = self._train_lines
        self._train_lines = []
        if not lines:
            return
        n_cal = max(1, int(len(lines) * self.config.calibration_fraction))
        fit_lines, cal_lines = lines[:-n_cal] or lines, lines[-n_cal:]
        tok = Tokenizer(models.BPE(unk_token=None))
        tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
        trainer = trainers.BpeTrainer(
            vocab_size=self.config.vocab_size, show_progress=False,
            initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
        )
        tok.train_from_iterator(fit_lines, trainer=trainer)
        self._tok = tok
        ratios = [self._ratio(l) for l in cal_lines]
        self._mean = sum(ratios) / len(ratios)
        var = sum((r - self._mean) ** 2 for r in ratios) / max(len(ratios), 1)
        self._std = max(math.sqrt(var), 1e-3)

    def _ratio(self, line: str) -> float:
        if not line:
            return 0.0
        return len(self._tok.encode(line).ids) / max(len(line.encode()), 1)

    # ------------------------------------------------------------------
    def train(self, parsed_batch: List[ParserSchema]) -> None:
        self._train_lines.extend(p.log or "" for p in parsed_batch)

    def detect(self, parsed: ParserSchema, alert: DetectorSchema) -> bool:
        if self._tok is None:
            if not self._train_lines:
                return False
            self._fit()
            if self._tok is None:
                return False
        z = (self._ratio(parsed.log or "") - self._mean) / self._std
        if z <= self.config.z_threshold:
            return False
        alert.detectorID = self.detector_id
        alert.detectorType = "tokenizer_detector"
        alert.alertID = f"tok-{parsed.logID}"
        alert.detectionTimestamp = int(time.time())
        alert.logIDs = [parsed.logID] if parsed.logID else []
        alert.score = float(z)
        alert.description = (
            f"Tokenization surprise: {z:.2f} sigma above the learned "
            f"tokens-per-byte baseline"
        )
        return True

    # -- checkpoint -----------------------------------------------------
    def state_dict(self) -> Dict[str, Any]:
        return {
            "seen_lines": self._seen_lines,
            "tokenizer_json": self._tok.to_str() if self._tok else None,
            "mean": self._mean,
            "std": self._std,
        }

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self._seen_lines = int(state.get("seen_lines", 0))
        self._mean = float(state.get("mean", 0.0))
        self._std = float(state.get("std", 1.0))
        tj = state.get("tokenizer_json")
        if tj:
            from tokenizers import Tokenizer

            self._tok = Tokenizer.from_str(tj)


class LLMAssistDetectorConfig(CoreDetectorConfig):
    method_type: str = "llm_assist_detector"
    #: inner detector method_type (resolved through the component system)
    inner: str = "new_value_detector"
    #: OpenAI-compatible chat completions base URL (a local vLLM /
    #: llama.cpp server); empty disables enrichment (pass-through)
    llm_base_url: str = ""
    llm_model: str = "local"
    llm_timeout_s: float = 3.0
    #: prompt template; {alert} and {log} are substituted
    llm_prompt: str = (
        "Summarize in one short sentence why this log line is anomalous.\n"
        "Alert: {alert}\nLine: {log}"
    )
    params: Dict[str, Any] = {}


class LLMAssistDetector(CoreDetector):
    """Alert enrichment through an OpenAI-compatible endpoint; detection
    itself is the wrapped inner detector's. LLM failures NEVER suppress
    an alert (assistance, not a dependency)."""

    CONFIG_CLASS = LLMAssistDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        from ..detectors import new_value as _nv  # default inner family
        from ...components.resolver import ComponentResolver
        from ...components.loader import ComponentLoader

        # resolve the inner detector by its class name through the same
        # component system the Service uses
        inner_name = self.config.inner
        if inner_name in ("new_value_detector", "NewValueDetector"):
            self.inner: CoreDetector = _nv.NewValueDetector(
                {**self.config.model_dump(exclude={"inner", "llm_base_url",
                                                   "llm_model", "llm_timeout_s",
                                                   "llm_prompt", "method_type"}),
                 "method_type": "new_value_detector"})
        else:
            path, _cfg = ComponentResolver().resolve(inner_name)
            self.inner = ComponentLoader().load_component(path, config=None)
        self.detector_id = f"llm_assist-{id(self):x}"

    def train(self, parsed_batch: List[ParserSchema]) -> None:
        self.inner.train(parsed_batch)

    def _enrich(self, alert: DetectorSchema, parsed: ParserSchema) -> None:
        base = self.config.llm_base_url
        if not base:
            return
        prompt = self.config.llm_prompt.format(
            alert=alert.description or "", log=parsed.log or "")
        body = json.dumps({
            "model": self.config.llm_model,
            "messages": [{"role": "user", "content": prompt}],
            "max_tokens": 64,
        }).encode()
        req = urllib.request.Request(
            base.rstrip("/") + "/v1/chat/completions", data=body,
            headers={"Content-Type": "application/json"})
        try:
            with urllib.request.urlopen(req, timeout=self.config.llm_timeout_s) as r:
                out = json.loads(r.read())
            text = out["choices"][0]["message"]["content"].strip()
            if text:
                alert.description = f"{alert.description} | LLM: {text}"
                if alert.alertsObtain is None:
                    alert.alertsObtain = {}
                alert.alertsObtain["llm_summary"] = text
        except Exception:  # noqa: BLE001 - enrichment is best-effort
            pass

    def detect(self, parsed: ParserSchema, alert: DetectorSchema) -> bool:
        hit = self.inner.detect(parsed, alert)
        if hit:
            self._enrich(alert, parsed)
        return hit

    def state_dict(self) -> Dict[str, Any]:
        return {"seen_lines": self._seen_lines,
                "inner": self.inner.state_dict()}

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self._seen_lines = int(state.get("seen_lines", 0))
        if state.get("inner"):
            self.inner.load_state_dict(state["inner"])
