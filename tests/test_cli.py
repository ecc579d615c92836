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


def test_client_cli_against_live_service(free_port, tmp_path, capsys):
    """detectmate-client subcommands against a running service."""
    import threading

    from detectmateservice_amd import Service, ServiceSettings
    from detectmateservice_amd.client import main as client_main

    settings = ServiceSettings(
        component_type="core", engine_addr=f"ipc://{tmp_path}/cli.ipc",
        http_port=free_port, log_dir=tmp_path / "logs",
        checkpoint_dir=tmp_path / "ckpts",
    )
    svc = Service(settings)
    t = threading.Thread(target=svc.run, daemon=True)
    t.start()
    try:
        assert svc.web_server.wait_started(10.0)
        url = f"http://127.0.0.1:{free_port}"
        assert client_main(["--url", url, "status"]) == 0
        out = capsys.readouterr().out
        assert '"running": true' in out
        assert client_main(["--url", url, "stop"]) == 0
        assert client_main(["--url", url, "start"]) == 0
        assert client_main(["--url", url, "metrics"]) == 0
        assert "engine_running" in capsys.readouterr().out
        cfg = tmp_path / "re.yaml"
        cfg.write_text("detectors: {}\n")
        assert client_main(["--url", url, "reconfigure", str(cfg)]) == 0
        # checkpoint names resolve under settings.checkpoint_dir
        assert client_main(["--url", url, "checkpoint", "c.pt"]) == 0
        assert (tmp_path / "ckpts" / "c.pt").exists()
        assert client_main(["--url", url, "restore", "c.pt"]) == 0
        assert client_main(["--url", url, "shutdown"]) == 0
    finally:
        svc.shutdown()
        t.join(timeout=10.0)


def test_service_logger_file_handler(tmp_path):
    from detectmateservice_amd.utils.logging import build_service_logger

    log = build_service_logger("testtype", "tid-1", "DEBUG", tmp_path / "lg")
    log.info("hello file")
    for h in log.handlers:
        h.flush()
    f = tmp_path / "lg" / "testtype_tid-1.log"
    assert f.exists() and "hello file" in f.read_text()
    # duplicate-handler guard: building again adds no handlers
    n = len(log.handlers)
    build_service_logger("testtype", "tid-1", "DEBUG", tmp_path / "lg")
    assert len(log.handlers) == n
