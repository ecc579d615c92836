"""MatcherParser: log_format field extraction + template matching.

Capability parity with the reference library's ``MatcherParser``
(``parsers.template_matcher``, method_type ``matcher_parser``; config shape
/root/reference/container/config/parser_config.yaml:1-12, usage
tests/library_integration/test_pipe_filereader_matcher_nvd.py:50-65):

* ``log_format`` — a token spec like
  ``'<IP> - - [<Time>] "<Method> <URL> <Protocol>" <Status> ...'``; each
  ``<Name>`` becomes a named header variable, the literal text between
  tokens must appear verbatim. Extracted fields land in
  ``ParserSchema.logFormatVariables``.
* templates — a file (``params.path_templates``) or inline list of
  patterns with ``<*>`` wildcards; the log content is matched against all
  templates, the first match yields ``EventID`` (1-based template index),
  ``template`` and the wildcard captures as ``variables``.
* normalization params ``remove_spaces`` / ``remove_punctuation`` /
  ``lowercase`` applied to the content before template matching.

The pure-Python matcher here is the semantic reference; the batched GPU
path (``detectmateservice_amd.ops.template_match``) runs the same
greedy-anchored wildcard match as a HIP kernel over an SoA byte batch and
is validated against this implementation in tests.
"""
from __future__ import annotations

import re
import string
import time
from pathlib import Path
from typing import Dict, List, Optional, Tuple

from ...components.base import CoreComponent, CoreConfig
from ...schemas import LogSchema, ParserSchema


class MatcherParserConfig(CoreConfig):
    method_type: str = "matcher_parser"
    #: mine templates from the first ``auto_config_lines`` lines instead of
    #: requiring a template file (reference parser_config.yaml:4 flag; the
    #: mining itself runs on the GPU edit-distance kernel — template_miner.py)
    auto_config: bool = False
    auto_config_lines: int = 1000
    #: drift adaptation: when > 0, buffer UNMATCHED (EventID -1) content
    #: lines and, once this many accumulate, mine them and APPEND any new
    #: templates (existing EventIDs are stable — new templates get higher
    #: ids). MI355X-first addition: the reference mines once at startup.
    auto_config_refit_lines: int = 0
    #: template aging: a template that matched NOTHING for this many
    #: observed lines is deactivated at the next drift refit (its
    #: EventID is retired, never reused — downstream per-event state
    #: stays valid). 0 disables aging.
    template_max_idle_lines: int = 0
    log_format: Optional[str] = None
    time_format: Optional[str] = None
    params: Dict = {}
    templates: List[str] = []


_TOKEN_RE = re.compile(r"<([A-Za-z_][A-Za-z0-9_]*)>")


def compile_log_format(log_format: str) -> re.Pattern:
    """``'<IP> - - [<Time>]'`` → anchored regex with named groups.

    Literal runs are escaped; each ``<Name>`` becomes a lazy ``(?P<Name>.+?)``
    except the last token, which is greedy so trailing fields absorb the
    rest of the line.
    """
    parts: List[str] = []
    pos = 0
    tokens = list(_TOKEN_RE.finditer(log_format))
    for i, m in enumerate(tokens):
        parts.append(re.escape(log_format[pos:m.start()]))
        greedy = ".+" if i == len(tokens) - 1 else ".+?"
        parts.append(f"(?P<{m.group(1)}>{greedy})")
        pos = m.end()
    parts.append(re.escape(log_format[pos:]))
    return re.compile("^" + "".join(parts) + "$")


def split_template(template: str) -> List[str]:
    """Template → literal segments (the text between ``<*>`` wildcards)."""
    return template.split("<*>")


def match_template(content: str, segments: List[str]) -> Optional[List[str]]:
    """Greedy-anchored wildcard match.

    Segments must appear in order; the first segment anchors at position 0
    when non-empty, the last must end the string when non-empty. Returns
    the wildcard captures, or None. This exact algorithm is mirrored by
    the HIP kernel in ops/csrc/template_match.hip.
    """
    captures: List[str] = []
    pos = 0
    n = len(segments)
    for i, seg in enumerate(segments):
        if seg == "":
            if i == n - 1:
                captures.append(content[pos:])
                return captures
            continue
        idx = content.find(seg, pos)
        if idx < 0:
            return None
        if i == 0 and idx != 0:
            return None
        if i > 0:
            captures.append(content[pos:idx])
        pos = idx + len(seg)
    if pos != len(content):
        if segments[-1] == "":
            pass  # trailing wildcard already captured
        else:
            return None
    return captures


_PUNCT_TABLE = str.maketrans("", "", string.punctuation)


class MatcherParser(CoreComponent):
    CONFIG_CLASS = MatcherParserConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        cfg = self.config
        self._format_re = (
            compile_log_format(cfg.log_format) if cfg.log_format else None
        )
        params = cfg.params or {}
        self._remove_spaces = bool(params.get("remove_spaces", False))
        self._remove_punct = bool(params.get("remove_punctuation", False))
        self._lowercase = bool(params.get("lowercase", False))
        templates = list(cfg.templates or [])
        tpl_path = params.get("path_templates")
        if tpl_path:
            p = Path(tpl_path)
            if p.exists():
                with open(p, "r", encoding="utf-8") as fh:
                    templates.extend(
                        line.rstrip("\n") for line in fh if line.strip()
                    )
        self.templates = templates
        self._segments = [split_template(t) for t in templates]
        #: aging state: EventIDs are positional and append-only, so aged
        #: templates DEACTIVATE (mask) instead of being removed
        self._active = [True] * len(templates)
        self._idle = [0] * len(templates)
        self.parser_id = f"matcher_parser-{id(self):x}"
        # auto_config: buffer content lines until the mining threshold,
        # then derive templates (template_miner.py) and start matching.
        self._auto_pending: Optional[List[str]] = (
            [] if (cfg.auto_config and not templates) else None
        )
        self._drift_pending: List[str] = []
        # Which extracted header field carries the free-text content that
        # template matching applies to. Defaults to the LAST token of the
        # log_format (e.g. ``<Content>`` in the audit format); when the
        # format does not match (or none is set) the whole line is matched.
        self._content_field = params.get("content_field")
        if self._content_field is None and cfg.log_format:
            tokens = _TOKEN_RE.findall(cfg.log_format)
            if tokens:
                self._content_field = tokens[-1]

    # -- normalization --------------------------------------------------
    def _normalize(self, content: str) -> str:
        if self._lowercase:
            content = content.lower()
        if self._remove_punct:
            content = content.translate(_PUNCT_TABLE)
        if self._remove_spaces:
            content = content.replace(" ", "")
        return content

    # -- auto_config mining --------------------------------------------
    def _auto_observe(self, lines: List[str]) -> None:
        """Collect content lines; once the threshold is reached, mine
        templates and switch to matching mode."""
        if self._auto_pending is None:
            return
        for line in lines:
            if self._format_re is not None:
                m = self._format_re.match(line)
                if m and self._content_field and self._content_field in m.groupdict():
                    line = m.group(self._content_field)
            self._auto_pending.append(self._normalize(line))
        if len(self._auto_pending) >= self.config.auto_config_lines:
            self.mine_templates()

    def _drift_observe(self, contents: List[str]) -> None:
        """Buffer unmatched content lines; refit when the threshold hits."""
        lim = self.config.auto_config_refit_lines
        if lim <= 0 or self._auto_pending is not None:
            return
        room = 4 * lim - len(self._drift_pending)
        if room > 0:
            self._drift_pending.extend(contents[:room])
        if len(self._drift_pending) >= lim:
            self.refit_templates()

    def _observe_usage(self, event_ids, batch_lines: int) -> None:
        """Aging bookkeeping: reset idle for templates that matched in
        this batch, advance it for the rest."""
        if self.config.template_max_idle_lines <= 0:
            return
        try:
            ids = {int(v) for v in event_ids.unique().tolist() if v > 0}
        except AttributeError:
            ids = {v for v in event_ids if v > 0}
        for i in range(len(self._idle)):
            if self._active[i]:
                self._idle[i] = 0 if (i + 1) in ids else self._idle[i] + batch_lines

    def refit_templates(self) -> List[str]:
        """Mine the drift buffer and append templates not already known.
        Existing templates keep their EventIDs (append-only)."""
        from .template_miner import TemplateMiner
        import torch

        # aging: retire templates that matched nothing for too long
        lim_idle = self.config.template_max_idle_lines
        if lim_idle > 0:
            aged = [i for i in range(len(self.templates))
                    if self._active[i] and self._idle[i] > lim_idle]
            for i in aged:
                self._active[i] = False
            if aged:
                self._batch_matcher = None  # rebuild without retired rows
        pending, self._drift_pending = self._drift_pending, []
        if not pending:
            return []
        device = (self.config.params or {}).get("device")
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        mined = TemplateMiner(device=device).fit(pending)
        known = set(self.templates)
        new = [m for m in mined if m not in known]
        if new:
            self.templates = self.templates + new
            self._segments = [split_template(t) for t in self.templates]
            self._active = self._active + [True] * len(new)
            self._idle = self._idle + [0] * len(new)
            self._batch_matcher = None  # rebuild packed tables
        return new

    def mine_templates(self) -> List[str]:
        """Run the miner on buffered lines and install the templates."""
        from .template_miner import TemplateMiner
        import torch

        pending = self._auto_pending or []
        self._auto_pending = None
        device = (self.config.params or {}).get("device")
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        miner = TemplateMiner(device=device)
        mined = miner.fit(pending)
        self.templates = mined
        self._segments = [split_template(t) for t in mined]
        self._active = [True] * len(mined)
        self._idle = [0] * len(mined)
        self._batch_matcher = None  # rebuild packed tables on next batch
        return mined

    # -- per-line parse (semantic reference for the GPU kernel) ---------
    def parse_line(self, line: str) -> Tuple[Dict[str, str], int, str, List[str]]:
        header: Dict[str, str] = {}
        content = line
        if self._format_re is not None:
            m = self._format_re.match(line)
            if m:
                header = {k: v for k, v in m.groupdict().items() if v is not None}
                if self._content_field and self._content_field in header:
                    content = header[self._content_field]
        content = self._normalize(content)
        for event_id, segments in enumerate(self._segments, start=1):
            if not self._active[event_id - 1]:
                continue
            caps = match_template(content, segments)
            if caps is not None:
                return header, event_id, self.templates[event_id - 1], caps
        return header, -1, "", []

    def process(self, data: bytes) -> Optional[bytes]:
        return self._process_python([data])[0]

    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]:
        """Batched path: C++ codec (decode N frames → SoA) + template-match
        kernel (GPU when available, CPU span matcher otherwise) + C++ batch
        encode. Falls back to the per-message Python path when the
        extension is absent or normalization params the kernel does not
        implement (remove_spaces / remove_punctuation) are set."""
        if self._auto_pending is not None:
            self._auto_observe(
                [LogSchema.deserialize(f).log for f in frames]
            )
        if (len(frames) >= 8 and self._remove_spaces is False
                and self._remove_punct is False and not self.config.time_format):
            try:
                return self._process_batched(frames)
            except RuntimeError:
                pass  # extension missing: python fallback
        return self._process_python(frames)

    def _extracted_timestamp(self, header: Dict[str, str]) -> Optional[int]:
        """Parse the header's time field with ``time_format`` (reference
        config key, container/config/parser_config.yaml:5). Field name
        defaults to "Time" (``params.time_field`` overrides)."""
        fmt = self.config.time_format
        if not fmt:
            return None
        field = (self.config.params or {}).get("time_field", "Time")
        value = header.get(field)
        if not value:
            return None
        import datetime

        try:
            dt = datetime.datetime.strptime(value, fmt)
            if dt.tzinfo is None:
                dt = dt.replace(tzinfo=datetime.timezone.utc)
            return int(dt.timestamp())
        except ValueError:
            return None

    def _process_python(self, frames: List[bytes]) -> List[Optional[bytes]]:
        now = int(time.time())
        out: List[Optional[bytes]] = []
        drift: List[str] = []
        used_ids: List[int] = []
        for raw in frames:
            log = LogSchema.deserialize(raw)
            header, event_id, template, variables = self.parse_line(log.log)
            used_ids.append(event_id)
            if event_id == -1 and self.config.auto_config_refit_lines > 0:
                content = log.log
                if self._format_re is not None:
                    m = self._format_re.match(content)
                    if m and self._content_field and self._content_field in m.groupdict():
                        content = m.group(self._content_field)
                drift.append(self._normalize(content))
            ts = self._extracted_timestamp(header)
            parsed = ParserSchema(
                parserType="matcher_parser",
                parserID=self.parser_id,
                EventID=event_id,
                template=template,
                variables=variables,
                logID=log.logID,
                log=log.log,
                logFormatVariables=header,
                receivedTimestamp=ts if ts is not None else now,
                parsedTimestamp=now,
            )
            out.append(parsed.serialize())
        self._observe_usage(used_ids, len(frames))
        if drift:
            self._drift_observe(drift)
        return out

    # -- batched SoA path ----------------------------------------------
    _BATCH_MAX_LEN = 512

    def _get_batch_matcher(self):
        if getattr(self, "_batch_matcher", None) is None:
            import torch

            from ... import ops

            if not ops.have_extension():
                raise RuntimeError("no extension")
            device = (self.config.params or {}).get("device")
            if device is None:
                device = "cuda" if torch.cuda.is_available() else "cpu"
            self._batch_device = device
            # aged templates are masked out; the matcher indexes only
            # ACTIVE templates and _ev_remap maps its 1-based ids back
            # to the stable positional EventIDs
            active_ids = [i for i in range(len(self.templates))
                          if self._active[i]]
            self._matcher_ids = active_ids
            self._ev_remap = torch.tensor(
                [0] + [i + 1 for i in active_ids], dtype=torch.int32)
            self._batch_matcher = ops.TemplateMatcher(
                [self.templates[i] for i in active_ids],
                log_format=self.config.log_format,
                lowercase=self._lowercase,
                device=device,
                max_len=self._BATCH_MAX_LEN,
            )
        return self._batch_matcher

    def _process_batched(self, frames: List[bytes]) -> List[Optional[bytes]]:
        from ... import ops
        from ...ops import _dmx_C  # type: ignore[attr-defined]

        matcher = self._get_batch_matcher()
        lines, lens, log_ids, _src, _host = _dmx_C.decode_log_batch(
            list(frames), self._BATCH_MAX_LEN
        )
        dev_lines = lines.to(self._batch_device)
        dev_lens = lens.to(self._batch_device)
        match = matcher.match_packed(dev_lines, dev_lens)
        match_cpu = {k: v.cpu() for k, v in match.items()}
        # remap matcher-local event ids to the stable positional ids
        ev_local = match_cpu["event_id"]
        if getattr(self, "_matcher_ids", None) is not None and (
                len(self._matcher_ids) != len(self.templates)):
            remapped = ev_local.clone()
            pos = ev_local > 0
            remapped[pos] = self._ev_remap[ev_local[pos].long()]
            match_cpu["event_id"] = remapped
        self._observe_usage(match_cpu["event_id"], len(frames))
        if self.config.auto_config_refit_lines > 0:
            ev = match_cpu["event_id"]
            idxs = (ev < 0).nonzero().flatten().tolist()[:256]
            if idxs:
                fc, nfc = match_cpu["fmt_caps"], match_cpu["n_fmt_caps"]
                drift = []
                for i in idxs:
                    row = bytes(lines[i, : int(lens[i])].numpy().tobytes())
                    nf = int(nfc[i])
                    if nf > 0:  # content = last header capture
                        s0, e0 = int(fc[i, nf - 1, 0]), int(fc[i, nf - 1, 1])
                        row = row[s0:e0]
                    drift.append(self._normalize(row.decode("utf-8", "replace")))
                self._drift_observe(drift)
        encoded = _dmx_C.encode_parser_batch(
            lines,
            lens,
            match_cpu["event_id"],
            match_cpu["caps"],
            match_cpu["n_caps"],
            match_cpu["fmt_caps"],
            match_cpu["n_fmt_caps"],
            list(matcher.fmt_field_names),
            list(self.templates),
            list(log_ids),
            "matcher_parser",
            self.parser_id,
            int(time.time()),
            "0.3",
        )
        return list(encoded)
