"""Template mining: discover ``<*>`` wildcard templates from raw lines.

Implements the reference family's ``auto_config`` capability (the
reference's MatcherParser config carries ``auto_config`` flags —
container/config/parser_config.yaml:4 — and its library depends on a
native Levenshtein extension for template mining).
Written fresh, MI355X-first:

1. lines are clustered greedily against cluster representatives by
   normalized Levenshtein distance — the pairwise distances run on the
   GPU wavefront-DP kernel (ops/csrc/edit_distance.hip) in batches, with
   a pure-Python fallback for CPU;
2. each cluster's members are token-aligned (whitespace); positions where
   all members agree stay literal, the rest become ``<*>`` (adjacent
   wildcards collapse).
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict, List, Optional, Sequence

import torch


def levenshtein_py(a: bytes, b: bytes) -> int:
    """CPU reference (also the GPU kernel's parity oracle in tests)."""
    m, n = len(a), len(b)
    if m == 0 or n == 0:
        return m + n
    prev = list(range(n + 1))
    for i in range(1, m + 1):
        cur = [i] + [0] * n
        ca = a[i - 1]
        for j in range(1, n + 1):
            cost = 0 if ca == b[j - 1] else 1
            cur[j] = min(prev[j] + 1, cur[j - 1] + 1, prev[j - 1] + cost)
        prev = cur
    return prev[n]


def edit_distances(
    queries: Sequence[bytes], refs: Sequence[bytes], device="cpu",
    max_len: int = 256,
) -> torch.Tensor:
    """[len(queries), len(refs)] int32 Levenshtein distances."""
    from .. import __name__ as _  # noqa: F401
    from ... import ops

    if len(queries) == 0 or len(refs) == 0:
        return torch.zeros((len(queries), len(refs)), dtype=torch.int32)
    dev = torch.device(device)
    if dev.type == "cuda" and ops.have_extension():
        from ...ops import _dmx_C  # type: ignore[attr-defined]

        qa, ql = ops.pack_lines(queries, max_len, device=dev)
        ra, rl = ops.pack_lines(refs, max_len, device=dev)
        return _dmx_C.edit_distance(qa, ql, ra, rl).cpu()
    out = torch.zeros((len(queries), len(refs)), dtype=torch.int32)
    for i, q in enumerate(queries):
        for j, r in enumerate(refs):
            out[i, j] = levenshtein_py(q[:max_len], r[:max_len])
    return out


class TemplateMiner:
    def __init__(
        self,
        max_norm_dist: float = 0.35,
        max_clusters: int = 512,
        min_cluster_size: int = 2,
        device: str | torch.device = "cpu",
        max_len: int = 256,
    ) -> None:
        self.max_norm_dist = max_norm_dist
        self.max_clusters = max_clusters
        self.min_cluster_size = min_cluster_size
        self.device = device
        self.max_len = max_len
        self.reps: List[bytes] = []
        self.members: Dict[int, List[str]] = defaultdict(list)

    # ------------------------------------------------------------------
    def add_lines(self, lines: Sequence[str], batch: int = 512) -> None:
        for off in range(0, len(lines), batch):
            chunk = [l for l in lines[off:off + batch] if l.strip()]
            if not chunk:
                continue
            raw = [l.encode("utf-8")[: self.max_len] for l in chunk]
            if self.reps:
                d = edit_distances(raw, self.reps, self.device, self.max_len)
            else:
                d = None
            n_pre = len(self.reps)  # reps covered by the batched distances
            new_in_chunk: List[int] = []
            for i, line in enumerate(chunk):
                assigned = None
                if d is not None and n_pre:
                    row = d[i, :n_pre]
                    j = int(torch.argmin(row))
                    norm = float(row[j]) / max(len(raw[i]), len(self.reps[j]), 1)
                    if norm <= self.max_norm_dist:
                        assigned = j
                if assigned is None:
                    # reps created within this chunk (not in the batched d)
                    for j in new_in_chunk:
                        ref = self.reps[j]
                        dd = levenshtein_py(raw[i], ref)
                        if dd / max(len(raw[i]), len(ref), 1) <= self.max_norm_dist:
                            assigned = j
                            break
                if assigned is None and len(self.reps) < self.max_clusters:
                    self.reps.append(raw[i])
                    assigned = len(self.reps) - 1
                    new_in_chunk.append(assigned)
                if assigned is not None:
                    self.members[assigned].append(line)

    # ------------------------------------------------------------------
    @staticmethod
    def _template_from(members: List[str]) -> Optional[str]:
        """Token-align members (majority token count) -> literal/<*> mix."""
        by_count: Dict[int, List[List[str]]] = defaultdict(list)
        for m in members:
            toks = m.split(" ")
            by_count[len(toks)].append(toks)
        count, group = max(by_count.items(), key=lambda kv: len(kv[1]))
        if count == 0:
            return None
        out: List[str] = []
        for pos in range(count):
            vals = {g[pos] for g in group}
            out.append(group[0][pos] if len(vals) == 1 else "<*>")
        # collapse adjacent wildcards
        collapsed: List[str] = []
        for tok in out:
            if tok == "<*>" and collapsed and collapsed[-1] == "<*>":
                continue
            collapsed.append(tok)
        tpl = " ".join(collapsed)
        return tpl if tpl.strip("<*> ") else None

    def templates(self) -> List[str]:
        seen = set()
        out: List[str] = []
        for cid in sorted(self.members, key=lambda c: -len(self.members[c])):
            if len(self.members[cid]) < self.min_cluster_size:
                continue
            tpl = self._template_from(self.members[cid])
            if tpl and tpl not in seen:
                seen.add(tpl)
                out.append(tpl)
        return out

    def fit(self, lines: Sequence[str]) -> List[str]:
        self.add_lines(list(lines))
        return self.templates()
