"""File reader: raw log source → LogSchema frames.

Capability parity with the reference library's ``readers.FileReader`` /
``readers.log_file.LogFileConfig`` (named in /root/reference/docs/
interfaces.md:98-99 and config_loader.py:22-24) and the ``From.log``
helper used by the reference's integration tests
(tests/library_integration/test_one_pipe_to_rule_them_all.py:22,136).
"""
from __future__ import annotations

import itertools
import uuid
from pathlib import Path
from typing import Iterator, List, Optional

from ...components.base import CoreComponent, CoreConfig
from ...schemas import LogSchema


class LogFileConfig(CoreConfig):
    path: Optional[str] = None
    log_source: str = "file"
    hostname: str = ""
    max_lines: Optional[int] = None
    #: source-mode streaming: keep tailing the file for appended lines
    follow: bool = False
    poll_interval_s: float = 0.2


class FileReaderConfig(LogFileConfig):
    pass


class FileReader(CoreComponent):
    """Reads a log file and emits one LogSchema frame per line.

    ``process(data)`` treats ``data`` as raw line bytes and wraps them;
    ``read()`` streams the configured file.
    """

    CONFIG_CLASS = FileReaderConfig

    def __init__(self, config=None) -> None:
        super().__init__(config)
        self._counter = itertools.count()

    def _wrap(self, line: str) -> LogSchema:
        return LogSchema(
            logID=f"{uuid.uuid4().hex[:12]}-{next(self._counter)}",
            log=line,
            logSource=self.config.log_source,
            hostname=self.config.hostname,
        )

    def process(self, data: bytes) -> Optional[bytes]:
        line = data.decode("utf-8", errors="replace").rstrip("\n")
        if not line:
            return None
        return self._wrap(line).serialize()

    def stream_batches(self, batch_size: int = 256, stop_event=None) -> Iterator[List[bytes]]:
        """Source-mode generator: serialized LogSchema frames in batches.

        With ``follow=True`` the reader tails the file (fluentd-style
        ingestion without fluentd); otherwise it ends at EOF. Used by the
        engine's source mode (``engine_source_mode: true``)."""
        import time as _time

        if not self.config.path:
            raise ValueError("FileReader requires config.path")
        limit = self.config.max_lines
        emitted = 0
        batch: List[bytes] = []
        with open(self.config.path, "r", encoding="utf-8", errors="replace") as fh:
            while stop_event is None or not stop_event.is_set():
                line = fh.readline()
                if not line:
                    if batch:
                        yield batch
                        batch = []
                    if not self.config.follow:
                        return
                    _time.sleep(self.config.poll_interval_s)
                    continue
                line = line.rstrip("\n")
                if not line:
                    continue
                batch.append(self._wrap(line).serialize())
                emitted += 1
                if limit is not None and emitted >= limit:
                    if batch:
                        yield batch
                    return
                if len(batch) >= batch_size:
                    yield batch
                    batch = []

    def read(self) -> Iterator[LogSchema]:
        if not self.config.path:
            raise ValueError("FileReader requires config.path")
        limit = self.config.max_lines
        with open(self.config.path, "r", encoding="utf-8", errors="replace") as fh:
            for i, line in enumerate(fh):
                if limit is not None and i >= limit:
                    break
                line = line.rstrip("\n")
                if line:
                    yield self._wrap(line)


class From:
    """Pipeline-in-a-call helpers mirroring the reference tests' ``From.log``."""

    @staticmethod
    def log(
        parser: Optional[CoreComponent],
        path: str | Path,
        do_process: bool = True,
        max_lines: Optional[int] = None,
    ) -> List[bytes]:
        """Stream a log file into LogSchema frames, optionally through a parser.

        Returns serialized LogSchema frames (``do_process=False`` or no
        parser) or the parser's output frames with None results dropped.
        """
        reader = FileReader({"path": str(path), "max_lines": max_lines})
        frames = [s.serialize() for s in reader.read()]
        if not do_process or parser is None:
            return frames
        outs = parser.process_batch(frames)
        return [o for o in outs if o is not None]
