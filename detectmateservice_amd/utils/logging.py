"""Logging helpers.

* :func:`setup_cli_logging` — root logging with <ERROR records on stdout and
  >=ERROR on stderr via a level filter (reference cli.py:12-32, tested in
  the reference's test_cli_logging_setup.py:21-53).
* :func:`build_service_logger` — per-component named logger
  ``{type}.{id}`` with console + per-component file handler
  ``logs/{type}_{id}.log``, no propagation, duplicate-handler guard
  (reference core.py:355-384).
"""
from __future__ import annotations

import logging
import sys
from pathlib import Path


class _BelowErrorFilter(logging.Filter):
    def filter(self, record: logging.LogRecord) -> bool:
        return record.levelno < logging.ERROR


def setup_cli_logging(level: str = "INFO") -> None:
    root = logging.getLogger()
    root.setLevel(level.upper())
    for h in list(root.handlers):
        root.removeHandler(h)
    fmt = logging.Formatter("%(asctime)s %(name)s %(levelname)s %(message)s")

    out = logging.StreamHandler(sys.stdout)
    out.addFilter(_BelowErrorFilter())
    out.setFormatter(fmt)
    root.addHandler(out)

    err = logging.StreamHandler(sys.stderr)
    err.setLevel(logging.ERROR)
    err.setFormatter(fmt)
    root.addHandler(err)


def build_service_logger(
    component_type: str,
    component_id: str,
    log_level: str = "INFO",
    log_dir: Path | str = "logs",
) -> logging.Logger:
    name = f"{component_type}.{component_id}"
    logger = logging.getLogger(name)
    logger.setLevel(log_level.upper())
    logger.propagate = False

    fmt = logging.Formatter("%(asctime)s %(name)s %(levelname)s %(message)s")
    have_console = any(
        isinstance(h, logging.StreamHandler) and not isinstance(h, logging.FileHandler)
        for h in logger.handlers
    )
    if not have_console:
        console = logging.StreamHandler(sys.stdout)
        console.setFormatter(fmt)
        logger.addHandler(console)

    log_dir = Path(log_dir)
    try:
        log_dir.mkdir(parents=True, exist_ok=True)
        file_path = log_dir / f"{component_type}_{component_id}.log"
        have_file = any(
            isinstance(h, logging.FileHandler)
            and getattr(h, "baseFilename", None) == str(file_path.resolve())
            for h in logger.handlers
        )
        if not have_file:
            fh = logging.FileHandler(file_path)
            fh.setFormatter(fmt)
            logger.addHandler(fh)
    except OSError:
        logger.warning("could not create log file in %s; console only", log_dir)
    return logger
