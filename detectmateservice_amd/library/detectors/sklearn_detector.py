"""SklearnDetector: classic-ML anomaly detection over per-line features.

The reference's library family depends on scikit-learn (SURVEY.md §2.2
capability hints — uv.lock lists scikit-learn among detectmatelibrary's
deps), i.e. classic-ML detectors exist in that family. This is the
MI355X framework's equivalent: a train-then-detect component fitting an
sklearn anomaly model (IsolationForest / LocalOutlierFactor /
OneClassSVM) on cheap numeric per-line features. Feature extraction is
vectorized numpy; the model itself is CPU (that is what sklearn is) —
the GPU path for learned detection is TransformerDetector /
EmbeddingDetector / the fused pipeline.
"""
from __future__ import annotations

import math
import time
from typing import Any, Dict, List, Optional

import numpy as np

from ...components.base import CoreDetector, CoreDetectorConfig
from ...schemas import DetectorSchema, ParserSchema

_PUNCT = frozenset(b"!\"#$%&'()*+,-./:;<=>?@[\\]^_`{|}~")


def line_features(lines: List[bytes]) -> np.ndarray:
    """[N, 8] float32: length, tokens, digit/alpha/punct fractions,
    byte entropy, mean token length, max token length."""
    out = np.zeros((len(lines), 8), dtype=np.float32)
    for i, raw in enumerate(lines):
        n = len(raw)
        if n == 0:
            continue
        toks = raw.split()
        counts = np.bincount(np.frombuffer(raw, dtype=np.uint8), minlength=256)
        p = counts[counts > 0] / n
        entropy = float(-(p * np.log2(p)).sum())
        n_digit = int(counts[48:58].sum())
        n_alpha = int(counts[65:91].sum() + counts[97:123].sum())
        n_punct = sum(counts[c] for c in _PUNCT)
        tok_lens = [len(t) for t in toks] or [0]
        out[i] = (
            n, len(toks), n_digit / n, n_alpha / n, n_punct / n,
            entropy, sum(tok_lens) / len(tok_lens), max(tok_lens),
        )
    return out


class SklearnDetectorConfig(CoreDetectorConfig):
    method_type: str = "sklearn_detector"
    params: Dict[str, Any] = {}


class SklearnDetector(CoreDetector):
    """``params``: ``model`` (isolation_forest | lof | one_class_svm),
    ``contamination`` (isolation_forest/lof), ``seed``.
    Trains on the first ``data_use_training`` lines; the model is fitted
    lazily on the first detect call after the training phase."""

    CONFIG_CLASS = SklearnDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        params = self.config.params or {}
        self.model_kind = str(params.get("model", "isolation_forest"))
        self.contamination = float(params.get("contamination", 0.01))
        self.seed = params.get("seed", 0)
        self._train_lines: List[bytes] = []
        self._model = None
        self.detector_id = f"sklearn_detector-{id(self):x}"

    # ------------------------------------------------------------------
    def _build_model(self):
        if self.model_kind == "isolation_forest":
            from sklearn.ensemble import IsolationForest

            return IsolationForest(
                n_estimators=100, contamination=self.contamination,
                random_state=self.seed,
            )
        if self.model_kind == "lof":
            from sklearn.neighbors import LocalOutlierFactor

            return LocalOutlierFactor(
                novelty=True, contamination=self.contamination
            )
        if self.model_kind == "one_class_svm":
            from sklearn.svm import OneClassSVM

            return OneClassSVM(nu=max(self.contamination, 1e-4), gamma="scale")
        raise ValueError(f"unknown sklearn model {self.model_kind!r}")

    def _fit_if_needed(self) -> None:
        if self._model is None and self._train_lines:
            x = line_features(self._train_lines)
            self._model = self._build_model().fit(x)
            self._train_lines = []

    # ------------------------------------------------------------------
    def train(self, parsed_batch: List[ParserSchema]) -> None:
        self._train_lines.extend(
            (p.log or "").encode("utf-8", "replace") for p in parsed_batch
        )

    def detect(self, parsed: ParserSchema, alert: DetectorSchema) -> bool:
        self._fit_if_needed()
        if self._model is None:
            return False
        x = line_features([(parsed.log or "").encode("utf-8", "replace")])
        pred = int(self._model.predict(x)[0])
        if pred != -1:
            return False
        score = float(-self._model.decision_function(x)[0])
        alert.detectorID = self.detector_id
        alert.detectorType = "sklearn_detector"
        alert.alertID = f"sk-{parsed.logID}"
        alert.detectionTimestamp = int(time.time())
        alert.logIDs = [parsed.logID] if parsed.logID else []
        alert.score = score
        alert.description = (
            f"{self.model_kind} anomaly (score {score:.4f})"
        )
        return True

    # -- checkpoint -----------------------------------------------------
    def state_dict(self) -> Dict[str, Any]:
        import pickle

        self._fit_if_needed()
        return {
            "model_kind": self.model_kind,
            "model_pickle": pickle.dumps(self._model) if self._model else None,
            "seen_lines": self._seen_lines,
        }

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        import pickle

        if state.get("model_pickle"):
            self._model = pickle.loads(state["model_pickle"])
        self._seen_lines = int(state.get("seen_lines", 0))


class FrequencyDetectorConfig(CoreDetectorConfig):
    method_type: str = "frequency_detector"
    #: lines per counting window
    window_lines: int = 1000
    #: EWMA smoothing for per-event window counts
    ewma_alpha: float = 0.3
    z_threshold: float = 4.0
    #: windows to observe before alerting (prime the EWMA)
    min_windows: int = 3
    params: Dict[str, Any] = {}


class FrequencyDetector(CoreDetector):
    """Event-rate anomaly: per-EventID counts in fixed line windows vs an
    EWMA baseline; a window whose count deviates ``z_threshold`` sigmas
    raises one alert on the boundary line. Catches floods and drop-outs
    that per-line detectors (NewValue, transformer scoring) cannot see.
    No reference equivalent (the reference family is per-line only)."""

    CONFIG_CLASS = FrequencyDetectorConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        self._counts: Dict[int, int] = {}
        self._mean: Dict[int, float] = {}
        self._var: Dict[int, float] = {}
        self._in_window = 0
        self._windows_seen = 0
        self.detector_id = f"frequency_detector-{id(self):x}"

    def _roll_window(self) -> List[str]:
        cfg = self.config
        anomalies: List[str] = []
        events = set(self._counts) | set(self._mean)
        for ev in events:
            c = float(self._counts.get(ev, 0))
            if ev in self._mean:
                mean, var = self._mean[ev], self._var[ev]
                std = math.sqrt(max(var, 1.0))
                z = (c - mean) / std
                if (abs(z) > cfg.z_threshold
                        and self._windows_seen >= cfg.min_windows):
                    anomalies.append(
                        f"event {ev}: {int(c)}/window vs {mean:.1f}±{std:.1f}"
                        f" (z={z:+.1f})"
                    )
                a = cfg.ewma_alpha
                self._mean[ev] = a * c + (1 - a) * mean
                self._var[ev] = a * (c - mean) ** 2 + (1 - a) * var
            else:
                self._mean[ev] = c
                self._var[ev] = max(c, 1.0)
        self._counts = {}
        self._in_window = 0
        self._windows_seen += 1
        return anomalies

    def train(self, parsed_batch: List[ParserSchema]) -> None:
        for p in parsed_batch:
            self._observe(p)

    def _observe(self, parsed: ParserSchema) -> List[str]:
        ev = int(parsed.EventID or 0)
        self._counts[ev] = self._counts.get(ev, 0) + 1
        self._in_window += 1
        if self._in_window >= self.config.window_lines:
            return self._roll_window()
        return []

    def detect(self, parsed: ParserSchema, alert: DetectorSchema) -> bool:
        anomalies = self._observe(parsed)
        if not anomalies:
            return False
        alert.detectorID = self.detector_id
        alert.detectorType = "frequency_detector"
        alert.alertID = f"freq-{parsed.logID}"
        alert.detectionTimestamp = int(time.time())
        alert.logIDs = [parsed.logID] if parsed.logID else []
        alert.score = float(len(anomalies))
        alert.description = "Rate anomaly: " + "; ".join(anomalies)
        return True

    # -- checkpoint -----------------------------------------------------
    def state_dict(self) -> Dict[str, Any]:
        return {
            "mean": dict(self._mean), "var": dict(self._var),
            "windows_seen": self._windows_seen,
            "seen_lines": self._seen_lines,
        }

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self._mean = {int(k): float(v) for k, v in state.get("mean", {}).items()}
        self._var = {int(k): float(v) for k, v in state.get("var", {}).items()}
        self._windows_seen = int(state.get("windows_seen", 0))
        self._seen_lines = int(state.get("seen_lines", 0))
