"""GPU ops layer: HIP kernels with CPU reference fallbacks.

Policy (build mandate): on a GPU box the HIP extension MUST load — any op
asked to run on a CUDA/HIP tensor raises if ``_dmx_C`` is missing (no
silent eager fallback). On CPU (CI container has no GPU) the pure-torch
reference paths run; they define the semantics the kernels are tested
against (tests/test_gpu_ops.py compares kernel output to these fp32
references).
"""
from __future__ import annotations

import math
from typing import List, Optional, Sequence, Tuple

import torch

_C = None
_C_ERR: Optional[str] = None
try:
    from . import _dmx_C as _C  # type: ignore[no-redef]
except ImportError as exc:  # pragma: no cover - exercised on GPU boxes
    _C_ERR = str(exc)


def have_extension() -> bool:
    return _C is not None


def _require_ext():
    if _C is None:
        raise RuntimeError(
            "detectmateservice_amd HIP extension (_dmx_C) is not built but a "
            "GPU tensor was passed. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_C_ERR}"
        )
    return _C


# ---------------------------------------------------------------------------
# dense ops
# ---------------------------------------------------------------------------

EPILOGUE = {"none": 0, "gelu": 1, "relu": 2}


def fused_linear(
    x: torch.Tensor,
    wt: torch.Tensor,
    bias: Optional[torch.Tensor] = None,
    activation: str = "none",
) -> torch.Tensor:
    """C = act(x @ wt.T + bias).  x [M,K] bf16; wt [N,K] bf16 (pre-transposed).

    GPU: hand-written MFMA kernel (ops/csrc/gemm_bf16.hip).
    CPU: torch reference in fp32 then cast back.
    """
    if x.is_cuda:
        return _require_ext().fused_linear_bf16(x, wt, bias, EPILOGUE[activation])
    y = torch.nn.functional.linear(x.float(), wt.float(), bias)
    if activation == "gelu":
        y = torch.nn.functional.gelu(y, approximate="tanh")
    elif activation == "relu":
        y = torch.relu(y)
    return y.to(x.dtype)


def layernorm(
    x: torch.Tensor,
    gamma: torch.Tensor,
    beta: torch.Tensor,
    residual: Optional[torch.Tensor] = None,
    eps: float = 1e-5,
    return_xres: bool = False,
):
    """y = LN(x [+ residual]) * gamma + beta (fused residual add).

    Optionally also returns x+residual (the next block's residual input).
    """
    if x.is_cuda:
        out = _require_ext().layernorm_bf16(x, residual, gamma, beta, eps, return_xres)
        return (out[0], out[1]) if return_xres else out[0]
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
    y = torch.nn.functional.layer_norm(xf, (x.shape[-1],), gamma.float(), beta.float(), eps)
    y = y.to(x.dtype)
    if return_xres:
        return y, xf.to(x.dtype)
    return y


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, scale: Optional[float] = None) -> torch.Tensor:
    """Fused MHA for [BH, S, Dh] with S<=128, Dh<=64 (bf16)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        return _require_ext().attention_bf16(q, k, v, scale)
    s = torch.softmax(q.float() @ k.float().transpose(-1, -2) * scale, dim=-1)
    return (s @ v.float()).to(q.dtype)


def attention_qkv(qkv: torch.Tensor, S: int, H: int, Dh: int, scale: Optional[float] = None) -> torch.Tensor:
    """MHA over the fused QKV projection output.

    qkv [B, S, 3*H*Dh] (q|k|v interleaved per row as the qkv linear emits
    them) -> O [B, S, H*Dh]. GPU: single MFMA kernel with no permutes
    (ops/csrc/attention_mfma.hip) when S%32==0, S<=128, Dh in {32,64};
    otherwise splits + falls back to :func:`attention`.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(Dh)
    B = qkv.shape[0]
    if qkv.is_cuda and S % 32 == 0 and S <= 128 and Dh in (32, 64):
        return _require_ext().attention_qkv_bf16(qkv, S, H, Dh, scale)
    # reference path: split + permute
    q, k, v = qkv.view(B, S, 3, H, Dh).unbind(dim=2)
    q = q.permute(0, 2, 1, 3).reshape(B * H, S, Dh).contiguous()
    k = k.permute(0, 2, 1, 3).reshape(B * H, S, Dh).contiguous()
    v = v.permute(0, 2, 1, 3).reshape(B * H, S, Dh).contiguous()
    o = attention(q, k, v, scale)
    return (
        o.view(B, H, S, Dh).permute(0, 2, 1, 3).reshape(B, S, H * Dh).contiguous()
    )


# ---------------------------------------------------------------------------
# parser: batched template matching
# ---------------------------------------------------------------------------


def pack_lines(lines: Sequence[bytes], max_len: int = 512, device="cpu") -> Tuple[torch.Tensor, torch.Tensor]:
    """Pad raw line bytes into a [B, max_len] u8 SoA tensor + lengths."""
    B = len(lines)
    buf = torch.zeros((B, max_len), dtype=torch.uint8)
    lens = torch.zeros((B,), dtype=torch.int32)
    for i, ln in enumerate(lines):
        b = ln[:max_len]
        if b:
            buf[i, : len(b)] = torch.frombuffer(bytearray(b), dtype=torch.uint8)
        lens[i] = len(b)
    return buf.to(device), lens.to(device)


def pack_templates(templates: Sequence[str]) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, int]:
    """Template strings -> (seg_bytes u8, seg_off i32, tpl_seg_start i32,
    max_caps). Segments are ``template.split('<*>')`` — identical to the
    Python matcher."""
    seg_bytes = bytearray()
    seg_off = [0]
    tpl_seg_start = [0]
    max_caps = 0
    for t in templates:
        segs = t.split("<*>")
        max_caps = max(max_caps, len(segs) - 1 + (1 if segs and segs[-1] == "" else 0))
        for s in segs:
            seg_bytes.extend(s.encode("utf-8"))
            seg_off.append(len(seg_bytes))
        tpl_seg_start.append(tpl_seg_start[-1] + len(segs))
    return (
        torch.frombuffer(bytearray(seg_bytes) or bytearray(b"\0"), dtype=torch.uint8).clone(),
        torch.tensor(seg_off, dtype=torch.int32),
        torch.tensor(tpl_seg_start, dtype=torch.int32),
        max(max_caps, 1),
    )


class TemplateMatcher:
    """Batched GPU/CPU template matcher sharing semantics with
    library/parsers/template_matcher.py (format split + content templates)."""

    def __init__(
        self,
        templates: Sequence[str],
        log_format: Optional[str] = None,
        lowercase: bool = False,
        device: str | torch.device = "cpu",
        max_len: int = 512,
    ) -> None:
        self.templates = list(templates)
        self.log_format = log_format
        self.lowercase = lowercase
        self.device = torch.device(device)
        self.max_len = max_len

        sb, so, ts, self.max_caps = pack_templates(self.templates)
        self.seg_bytes = sb.to(self.device)
        self.seg_off = so.to(self.device)
        self.tpl_seg_start = ts.to(self.device)

        if log_format:
            import re

            fmt_tpl = re.sub(r"<[A-Za-z_][A-Za-z0-9_]*>", "<*>", log_format)
            self.fmt_field_names = re.findall(r"<([A-Za-z_][A-Za-z0-9_]*)>", log_format)
            fb, fo, _fts, self.max_fmt_caps = pack_templates([fmt_tpl])
            self.fmt_bytes = fb.to(self.device)
            self.fmt_seg_off = fo.to(self.device)
        else:
            self.fmt_field_names = []
            self.fmt_bytes = torch.zeros(0, dtype=torch.uint8, device=self.device)
            self.fmt_seg_off = torch.zeros(0, dtype=torch.int32, device=self.device)
            self.max_fmt_caps = 1

    def match_packed(self, lines: torch.Tensor, line_len: torch.Tensor):
        """lines [B, max_len] u8 on self.device. Returns dict of tensors:
        event_id [B], fmt_caps [B,Fc,2], n_fmt_caps [B], caps [B,C,2], n_caps [B]."""
        if lines.shape[0] == 0:  # zero-grid kernel launches are invalid
            z = lambda *shape: torch.zeros(shape, dtype=torch.int32, device=lines.device)  # noqa: E731
            return {"event_id": z(0), "fmt_caps": z(0, self.max_fmt_caps, 2),
                    "n_fmt_caps": z(0), "caps": z(0, self.max_caps, 2),
                    "n_caps": z(0), "span_start": z(0), "span_end": z(0)}
        if lines.is_cuda:
            (ev, fc, nfc, caps, ncaps, sps, spe) = _require_ext().template_match(
                lines, line_len, self.fmt_bytes, self.fmt_seg_off,
                self.seg_bytes, self.seg_off, self.tpl_seg_start,
                self.lowercase, self.max_fmt_caps, self.max_caps,
            )
            return {"event_id": ev, "fmt_caps": fc, "n_fmt_caps": nfc,
                    "caps": caps, "n_caps": ncaps, "span_start": sps,
                    "span_end": spe}
        if _C is not None:
            # C++ twin of the kernel (the pure-Python matcher measured
            # ~9.5k lines/s and bottlenecked CPU service mode)
            ev, fc, nfc, caps, ncaps = _C.template_match_cpu(
                lines, line_len, self.fmt_bytes, self.fmt_seg_off,
                self.seg_bytes, self.seg_off, self.tpl_seg_start,
                self.lowercase, self.max_fmt_caps, self.max_caps,
            )
            out = {"event_id": ev, "fmt_caps": fc, "n_fmt_caps": nfc,
                   "caps": caps, "n_caps": ncaps}
            self._add_spans(out, line_len)
            return out
        out = self._match_cpu(lines, line_len)
        self._add_spans(out, line_len)
        return out

    @staticmethod
    def _add_spans(match: dict, line_len: torch.Tensor) -> None:
        """CPU-path twin of the kernel's span outputs: content span =
        last header capture when the format matched, else [0, len)."""
        nfc = match["n_fmt_caps"].long()
        has_hdr = nfc > 0
        last = (nfc - 1).clamp(min=0)
        fc = match["fmt_caps"]
        start = torch.where(
            has_hdr,
            fc.gather(1, last.view(-1, 1, 1).expand(-1, 1, 2))[:, 0, 0].long(),
            torch.zeros_like(nfc),
        )
        end = torch.where(
            has_hdr,
            fc.gather(1, last.view(-1, 1, 1).expand(-1, 1, 2))[:, 0, 1].long(),
            line_len.long(),
        )
        match["span_start"] = start.int()
        match["span_end"] = end.int()

    # -- CPU reference (same algorithm, used for CI + kernel parity tests) --
    def _match_cpu(self, lines: torch.Tensor, line_len: torch.Tensor):
        from ..library.parsers.template_matcher import match_template

        B = lines.shape[0]
        ev = torch.full((B,), -1, dtype=torch.int32)
        fc = torch.zeros((B, self.max_fmt_caps, 2), dtype=torch.int32)
        nfc = torch.zeros((B,), dtype=torch.int32)
        caps = torch.zeros((B, self.max_caps, 2), dtype=torch.int32)
        ncaps = torch.zeros((B,), dtype=torch.int32)

        fmt_segs = None
        if self.log_format:
            import re

            fmt_segs = re.sub(
                r"<[A-Za-z_][A-Za-z0-9_]*>", "<*>", self.log_format
            ).split("<*>")
        tpl_segs = [t.split("<*>") for t in self.templates]

        for i in range(B):
            raw = bytes(lines[i, : int(line_len[i])].numpy().tobytes())
            text = raw.decode("utf-8", errors="replace")
            start, end = 0, len(text)
            if fmt_segs:
                spans = _span_match(text, 0, len(text), fmt_segs)
                if spans is not None:
                    nfc[i] = len(spans)
                    for j, (a, b) in enumerate(spans[: self.max_fmt_caps]):
                        fc[i, j, 0], fc[i, j, 1] = a, b
                    if spans:
                        start, end = spans[-1]
            cmp_text = text.lower() if self.lowercase else text
            for t, segs in enumerate(tpl_segs):
                spans = _span_match(cmp_text, start, end, segs)
                if spans is not None:
                    ev[i] = t + 1
                    ncaps[i] = len(spans)
                    for j, (a, b) in enumerate(spans[: self.max_caps]):
                        caps[i, j, 0], caps[i, j, 1] = a, b
                    break
        return {"event_id": ev, "fmt_caps": fc, "n_fmt_caps": nfc,
                "caps": caps, "n_caps": ncaps}


def _span_match(text: str, start: int, end: int, segments: List[str]):
    """match_template returning (start, end) spans instead of strings —
    the exact algorithm of template_match.hip::match_segments."""
    pos = start
    spans: List[Tuple[int, int]] = []
    n = len(segments)
    for i, seg in enumerate(segments):
        if seg == "":
            if i == n - 1:
                spans.append((pos, end))
                return spans
            continue
        idx = text.find(seg, pos, end)
        if idx < 0:
            return None
        if i == 0 and idx != start:
            return None
        if i > 0:
            spans.append((pos, idx))
        pos = idx + len(seg)
    if pos != end:
        return None
    return spans


# ---------------------------------------------------------------------------
# detector: GPU hash sets
# ---------------------------------------------------------------------------


class GpuHashSets:
    """W open-addressing u64 hash sets (one per watched field) resident on
    device; CPU fallback keeps Python sets with the same FNV-1a hashing."""

    def __init__(self, n_watch: int, capacity: int = 1 << 16, device="cpu") -> None:
        assert capacity & (capacity - 1) == 0
        self.W = n_watch
        self.capacity = capacity
        self.device = torch.device(device)
        if self.device.type == "cuda":
            self.tables = torch.zeros((n_watch, capacity), dtype=torch.int64, device=self.device)
        else:
            self.sets: List[set] = [set() for _ in range(n_watch)]

    def insert(self, hashes: torch.Tensor) -> None:
        if hashes.shape[0] == 0:
            return
        if self.device.type == "cuda":
            _require_ext().hashset_insert(hashes, self.tables)
        else:
            h = hashes.cpu().numpy()
            for w in range(self.W):
                self.sets[w].update(int(x) for x in h[:, w] if x != 0)

    def probe(self, hashes: torch.Tensor) -> torch.Tensor:
        if hashes.shape[0] == 0:
            return torch.zeros((0, self.W), dtype=torch.int32, device=hashes.device)
        if self.device.type == "cuda":
            return _require_ext().hashset_probe(hashes, self.tables)
        h = hashes.cpu().numpy()
        out = torch.zeros(hashes.shape, dtype=torch.int32)
        for w in range(self.W):
            s = self.sets[w]
            for i in range(h.shape[0]):
                v = int(h[i, w])
                out[i, w] = 1 if (v != 0 and v not in s) else 0
        return out

    def state_dict(self):
        if self.device.type == "cuda":
            nz = self.tables[self.tables != 0]
            return {"type": "gpu", "tables": self.tables.cpu()}
        return {"type": "cpu", "sets": [sorted(s) for s in self.sets]}

    def load_state_dict(self, state):
        if self.device.type == "cuda":
            if state["type"] == "gpu":
                self.tables.copy_(state["tables"].to(self.device))
            else:
                for w, vals in enumerate(state["sets"]):
                    if vals:
                        h = torch.zeros((len(vals), self.W), dtype=torch.int64)
                        h[:, w] = torch.tensor(vals, dtype=torch.int64)
                        self.insert(h.to(self.device))
        else:
            if state["type"] == "cpu":
                self.sets = [set(v) for v in state["sets"]]
            else:
                t = state["tables"]
                for w in range(self.W):
                    self.sets[w] = set(int(x) for x in t[w][t[w] != 0].tolist())


def fnv1a64(data: bytes, lower: bool = False) -> int:
    """CPU mirror of the kernel's hash (for parity tests + CPU fallback)."""
    h = 1469598103934665603
    for c in data:
        if lower and 65 <= c <= 90:
            c += 32
        h ^= c
        h = (h * 1099511628211) & 0xFFFFFFFFFFFFFFFF
    return h | 1


def watch_hashes_cpu(
    lines: torch.Tensor,
    match: dict,
    specs: torch.Tensor,
    lower: bool = False,
) -> torch.Tensor:
    """CPU mirror of dmx_watch_hashes."""
    B = lines.shape[0]
    W = specs.shape[0]
    out = torch.zeros((B, W), dtype=torch.int64)
    ev = match["event_id"]
    caps, ncaps = match["caps"], match["n_caps"]
    fcaps, nfcaps = match["fmt_caps"], match["n_fmt_caps"]
    for i in range(B):
        row = bytes(lines[i].numpy().tobytes())
        for w in range(W):
            kind, event, pos = int(specs[w, 0]), int(specs[w, 1]), int(specs[w, 2])
            start = end = -1
            if kind == 0:
                if (event < 0 or int(ev[i]) == event) and pos < int(ncaps[i]):
                    start, end = int(caps[i, pos, 0]), int(caps[i, pos, 1])
            else:
                if pos < int(nfcaps[i]):
                    start, end = int(fcaps[i, pos, 0]), int(fcaps[i, pos, 1])
            if start >= 0 and end >= start:
                h = fnv1a64(row[start:end], lower)
                out[i, w] = h - (1 << 64) if h >= (1 << 63) else h
    return out


def watch_hashes(lines, match: dict, specs: torch.Tensor, lower: bool = False) -> torch.Tensor:
    if lines.is_cuda:
        return _require_ext().watch_hashes(
            lines, match["event_id"], match["caps"], match["n_caps"],
            match["fmt_caps"], match["n_fmt_caps"], specs, lower,
        )
    return watch_hashes_cpu(lines, match, specs, lower)
