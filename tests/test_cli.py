"""CLI tests: logging split, settings requirement, client command mapping
(reference test_cli_logging_setup.py:21-53 shape)."""
import logging

import pytest

from detectmateservice_amd.cli import main as cli_main
from detectmateservice_amd.utils.logging import setup_cli_logging


def test_logging_split_stdout_stderr(capsys):
    setup_cli_logging("DEBUG")
    log = logging.getLogger("split-test")
    log.info("info goes to stdout")
    log.error("error goes to stderr")
    captured = capsys.readouterr()
    assert "info goes to stdout" in captured.out
    assert "info goes to stdout" not in captured.err
    assert "error goes to stderr" in captured.err
    assert "error goes to stderr" not in captured.out


def test_cli_requires_settings(capsys):
    rc = cli_main([])
    assert rc == 2
    assert "required" in capsys.readouterr().err


def test_cli_bad_settings_file(capsys, tmp_path):
    missing = tmp_path / "nope.yaml"
    rc = cli_main(["--settings", str(missing)])
    assert rc == 2


def test_client_parser():
    from detectmateservice_amd.client import main as client_main

    with pytest.raises(SystemExit):
        client_main([])  # subcommand required
