"""Multi-process pipeline integration: real service subprocesses chained
over ipc sockets (the reference's library_integration tier,
tests/library_integration/library_integration_base.py:12-53 shape).

reader(test driver) → parser(MatcherParser) → detector(NewValueDetector)
→ sink(test listener): LogSchema in, DetectorSchema alert out.
"""
import signal
import subprocess
import sys
import time
import uuid

import httpx
import pytest
import yaml

from detectmateservice_amd.engine.sockets import PairDialer, PairListener, RecvTimeout
from detectmateservice_amd.schemas import DetectorSchema, LogSchema, ParserSchema
from detectmateservice_amd.utils.synthetic import (
    AUDIT_LOG_FORMAT,
    AUDIT_TEMPLATES,
    AuditLogGenerator,
)


def _wait_running(port: int, timeout_s: float = 30.0) -> bool:
    deadline = time.monotonic() + timeout_s
    while time.monotonic() < deadline:
        try:
            r = httpx.get(f"http://127.0.0.1:{port}/admin/status", timeout=2.0)
            if r.status_code == 200 and r.json()["status"]["running"]:
                return True
        except Exception:  # noqa: BLE001
            pass
        time.sleep(0.2)
    return False


@pytest.fixture
def pipeline_procs(tmp_path, free_port):
    """Spawn parser + detector service subprocesses; yield addresses."""
    uid = uuid.uuid4().hex[:6]
    parser_in = f"ipc://{tmp_path}/parser-{uid}.ipc"
    detector_in = f"ipc://{tmp_path}/detector-{uid}.ipc"
    sink_addr = f"ipc://{tmp_path}/sink-{uid}.ipc"
    parser_port = free_port
    import socket as s

    sock = s.socket()
    sock.bind(("127.0.0.1", 0))
    detector_port = sock.getsockname()[1]
    sock.close()

    tpl_file = tmp_path / "templates.txt"
    tpl_file.write_text("\n".join(AUDIT_TEMPLATES) + "\n")

    parser_settings = tmp_path / "parser_settings.yaml"
    parser_settings.write_text(yaml.safe_dump({
        "component_type": "MatcherParser",
        "component_name": f"parser-{uid}",
        "engine_addr": parser_in,
        "out_addr": [detector_in],
        "http_port": parser_port,
        "config_file": str(tmp_path / "parser_config.yaml"),
        "log_dir": str(tmp_path / "logs"),
        "engine_batch_linger_ms": 5.0,
    }))
    (tmp_path / "parser_config.yaml").write_text(yaml.safe_dump({
        "parsers": {"MatcherParser": {
            "method_type": "matcher_parser",
            "log_format": AUDIT_LOG_FORMAT,
            "params": {"path_templates": str(tpl_file)},
        }}
    }))

    detector_settings = tmp_path / "detector_settings.yaml"
    detector_settings.write_text(yaml.safe_dump({
        "component_type": "NewValueDetector",
        "component_name": f"detector-{uid}",
        "engine_addr": detector_in,
        "out_addr": [sink_addr],
        "http_port": detector_port,
        "config_file": str(tmp_path / "detector_config.yaml"),
        "log_dir": str(tmp_path / "logs"),
        "engine_batch_linger_ms": 5.0,
    }))
    (tmp_path / "detector_config.yaml").write_text(yaml.safe_dump({
        "detectors": {"NewValueDetector": {
            "method_type": "new_value_detector",
            "data_use_training": 4,
            "global": {"g": {"header_variables": [{"pos": "Type"}]}},
        }}
    }))

    procs = []
    for settings in (parser_settings, detector_settings):
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "detectmateservice_amd.cli",
             "--settings", str(settings)],
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        ))
    try:
        assert _wait_running(parser_port), "parser service did not start"
        assert _wait_running(detector_port), "detector service did not start"
        yield {
            "parser_in": parser_in,
            "sink": sink_addr,
            "parser_port": parser_port,
            "detector_port": detector_port,
        }
    finally:
        for port in (parser_port, detector_port):
            try:
                httpx.post(f"http://127.0.0.1:{port}/admin/shutdown", timeout=2.0)
            except Exception:  # noqa: BLE001
                pass
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.send_signal(signal.SIGINT)
                try:
                    p.wait(timeout=5)
                except subprocess.TimeoutExpired:
                    p.kill()


def test_three_stage_pipeline(pipeline_procs):
    """Known Type values trained; an unseen Type raises an alert at the
    sink; normal lines produce nothing (downstream timeout)."""
    sink = PairListener(pipeline_procs["sink"])
    feeder = PairDialer(pipeline_procs["parser_in"])
    try:
        assert feeder.wait_connected(10.0)
        gen = AuditLogGenerator(seed=55)

        # train: 4 lines with Type values the detector learns
        train_lines = []
        while len(train_lines) < 4:
            line, _, _ = gen.line()
            if line.startswith("type=USER_ACCT") or line.startswith("type=LOGIN"):
                train_lines.append(line)
        for i, line in enumerate(train_lines):
            feeder.send(LogSchema(logID=f"train{i}", log=line).serialize())

        time.sleep(1.0)  # allow training frames through both stages

        # normal: same Type -> no alert
        feeder.send(LogSchema(logID="ok", log=train_lines[0]).serialize())
        with pytest.raises(RecvTimeout):
            sink.recv(timeout_ms=1500)

        # anomaly: a Type never seen in training
        bad = train_lines[0].replace(
            train_lines[0].split(" ", 1)[0], "type=EVIL_EVENT"
        )
        feeder.send(LogSchema(logID="bad1", log=bad).serialize())
        alert = DetectorSchema.deserialize(sink.recv(timeout_ms=10000))
        assert alert.detectorType == "new_value_detector"
        assert "EVIL_EVENT" in alert.description
    finally:
        feeder.close()
        sink.close()


def test_four_stage_with_output_aggregator(tmp_path, free_port):
    """parser -> detector -> OutputAggregator services; OutputSchema frames
    arrive at the final sink (the fluentout-equivalent egress)."""
    import socket as s_mod
    import uuid as uuid_mod

    from detectmateservice_amd.schemas import OutputSchema

    uid = uuid_mod.uuid4().hex[:6]
    det_in = f"ipc://{tmp_path}/d4-{uid}.ipc"
    out_in = f"ipc://{tmp_path}/o4-{uid}.ipc"
    sink_addr = f"ipc://{tmp_path}/s4-{uid}.ipc"
    sock = s_mod.socket(); sock.bind(("127.0.0.1", 0))
    out_port = sock.getsockname()[1]; sock.close()

    det_settings = tmp_path / "d4.yaml"
    det_settings.write_text(yaml.safe_dump({
        "component_type": "NewValueDetector",
        "engine_addr": det_in,
        "out_addr": [out_in],
        "http_port": free_port,
        "config_file": str(tmp_path / "d4c.yaml"),
        "log_dir": str(tmp_path / "logs"),
    }))
    (tmp_path / "d4c.yaml").write_text(yaml.safe_dump({
        "detectors": {"NewValueDetector": {
            "data_use_training": 2,
            "global": {"g": {"header_variables": [{"pos": "Type"}]}},
        }}
    }))
    agg_settings = tmp_path / "a4.yaml"
    agg_settings.write_text(yaml.safe_dump({
        "component_type": "OutputAggregator",
        "engine_addr": out_in,
        "out_addr": [sink_addr],
        "http_port": out_port,
        "config_file": str(tmp_path / "a4c.yaml"),
        "log_dir": str(tmp_path / "logs"),
    }))
    (tmp_path / "a4c.yaml").write_text(yaml.safe_dump({
        "outputs": {"OutputAggregator": {"window_size": 1}}
    }))

    procs = [subprocess.Popen(
        [sys.executable, "-m", "detectmateservice_amd.cli", "--settings", str(st)],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    ) for st in (det_settings, agg_settings)]
    sink = PairListener(sink_addr)
    feeder = PairDialer(det_in)
    try:
        assert _wait_running(free_port)
        assert _wait_running(out_port)
        assert feeder.wait_connected(10.0)

        def pframe(t, lid):
            return ParserSchema(
                EventID=1, logID=lid, logFormatVariables={"Type": t}
            ).serialize()

        feeder.send(pframe("LOGIN", "t1"))
        feeder.send(pframe("USER_ACCT", "t2"))
        time.sleep(0.8)
        feeder.send(pframe("WEIRD_TYPE", "bad"))
        out = OutputSchema.deserialize(sink.recv(timeout_ms=15000))
        assert out.detectorTypes == ["new_value_detector"]
        assert out.logIDs == ["bad"]
        assert "WEIRD_TYPE" in out.description
    finally:
        feeder.close()
        sink.close()
        for port in (free_port, out_port):
            try:
                httpx.post(f"http://127.0.0.1:{port}/admin/shutdown", timeout=2.0)
            except Exception:  # noqa: BLE001
                pass
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()


def test_pipeline_metrics_flow(pipeline_procs):
    """Prometheus counters advance along the pipeline."""
    feeder = PairDialer(pipeline_procs["parser_in"])
    try:
        assert feeder.wait_connected(10.0)
        gen = AuditLogGenerator(seed=66)
        for i in range(10):
            feeder.send(LogSchema(logID=f"m{i}", log=gen.line()[0]).serialize())
        deadline = time.monotonic() + 15
        ok = False
        while time.monotonic() < deadline and not ok:
            m = httpx.get(
                f"http://127.0.0.1:{pipeline_procs['parser_port']}/metrics",
                timeout=5.0,
            ).text
            for ln in m.splitlines():
                if ln.startswith("data_processed_lines_total") and float(ln.rsplit(" ", 1)[1]) >= 10:
                    ok = True
            time.sleep(0.3)
        assert ok, "parser did not count 10 processed lines"
    finally:
        feeder.close()
