"""Service-level integration: service in a daemon thread, admin over real
HTTP (reference shapes: tests/test_smoke_service.py, test_engine_loop.py
service tier)."""
import threading
import time

import httpx
import pytest

from detectmateservice_amd import Service, ServiceSettings
from detectmateservice_amd.engine.sockets import PairDialer, RecvTimeout
from detectmateservice_amd.schemas import DetectorSchema, LogSchema, ParserSchema


@pytest.fixture
def running_service(ipc_addr, free_port, tmp_path):
    """A NewValueDetector-less echo service with live HTTP admin."""
    settings = ServiceSettings(
        component_type="core",
        component_name="svc-under-test",
        engine_addr=ipc_addr,
        http_host="127.0.0.1",
        http_port=free_port,
        config_file=tmp_path / "cfg.yaml",
        log_dir=tmp_path / "logs",
    )
    service = Service(settings)
    t = threading.Thread(target=service.run, daemon=True)
    t.start()
    assert service.web_server.wait_started(10.0)
    yield service, f"http://127.0.0.1:{free_port}"
    service.shutdown()
    t.join(timeout=5.0)


def test_status_and_metrics_endpoints(running_service):
    service, url = running_service
    r = httpx.get(f"{url}/admin/status", timeout=5.0)
    assert r.status_code == 200
    body = r.json()
    assert body["status"]["running"] is True
    assert body["status"]["engine_running"] is True
    assert body["settings"]["component_name"] == "svc-under-test"

    m = httpx.get(f"{url}/metrics", timeout=5.0)
    assert m.status_code == 200
    assert "engine_running" in m.text
    assert "processing_duration_seconds" in m.text


def test_stop_start_via_http(running_service):
    service, url = running_service
    assert httpx.post(f"{url}/admin/stop", timeout=5.0).json()["status"] == "stopped"
    assert not service.engine.running
    assert httpx.post(f"{url}/admin/start", timeout=5.0).json()["status"] == "started"
    assert service.engine.running


def test_reconfigure_via_http(running_service, tmp_path):
    service, url = running_service
    payload = {
        "config": {"detectors": {"NewValueDetector": {"data_use_training": 3}}},
        "persist": True,
    }
    r = httpx.post(f"{url}/admin/reconfigure", json=payload, timeout=5.0)
    assert r.status_code == 200
    assert (
        service.config_manager.component_section("NewValueDetector")[
            "data_use_training"
        ]
        == 3
    )
    # persisted to the config file
    assert service.settings.config_file.exists()
    text = service.settings.config_file.read_text()
    assert "data_use_training" in text


def test_echo_processing_through_service(running_service):
    _service, url = running_service
    service, _ = running_service
    client = PairDialer(service.settings.engine_addr)
    try:
        assert client.wait_connected(5.0)
        client.send(b"raw-bytes")
        assert client.recv(timeout_ms=3000) == b"raw-bytes"  # passthrough (core type)
    finally:
        client.close()


def test_shutdown_via_http(ipc_addr, free_port, tmp_path):
    settings = ServiceSettings(
        component_type="core",
        engine_addr=ipc_addr,
        http_port=free_port,
        log_dir=tmp_path / "logs",
    )
    service = Service(settings)
    t = threading.Thread(target=service.run, daemon=True)
    t.start()
    assert service.web_server.wait_started(10.0)
    r = httpx.post(f"http://127.0.0.1:{free_port}/admin/shutdown", timeout=5.0)
    assert r.status_code == 200
    t.join(timeout=10.0)
    assert not t.is_alive()


def test_detector_service_pipeline(ipc_addr, tmp_path, free_port):
    """A NewValueDetector service: train 2 frames, alert on new URL
    (reference library-integration shape, single process)."""
    import yaml

    cfg_file = tmp_path / "detector.yaml"
    cfg_file.write_text(
        yaml.safe_dump(
            {
                "detectors": {
                    "NewValueDetector": {
                        "method_type": "new_value_detector",
                        "data_use_training": 2,
                        "global": {
                            "g": {"header_variables": [{"pos": "URL"}]}
                        },
                    }
                }
            }
        )
    )
    settings = ServiceSettings(
        component_type="NewValueDetector",
        engine_addr=ipc_addr,
        config_file=cfg_file,
        http_enabled=False,
        log_dir=tmp_path / "logs",
        engine_batch_linger_ms=0.0,
    )
    service = Service(settings)
    t = threading.Thread(target=service.run, daemon=True)
    t.start()
    time.sleep(0.2)

    def parsed_frame(url, log_id):
        return ParserSchema(
            EventID=1, logID=log_id, logFormatVariables={"URL": url}
        ).serialize()

    client = PairDialer(ipc_addr)
    try:
        assert client.wait_connected(5.0)
        client.send(parsed_frame("/a", "t1"))
        client.send(parsed_frame("/b", "t2"))
        client.send(parsed_frame("/a", "ok"))  # known → no alert
        with pytest.raises(RecvTimeout):
            client.recv(timeout_ms=400)
        client.send(parsed_frame("/foobar", "bad"))
        alert = DetectorSchema.deserialize(client.recv(timeout_ms=5000))
        assert alert.description == "Unknown value: '/foobar'"
        assert alert.logIDs == ["bad"]
    finally:
        client.close()
        service.shutdown()
        t.join(timeout=5.0)


def test_status_includes_live_counters(running_service):
    service, url = running_service
    client = PairDialer(service.settings.engine_addr)
    try:
        assert client.wait_connected(5.0)
        client.send(b"count-me")
        assert client.recv(timeout_ms=3000) == b"count-me"
        deadline = time.time() + 5
        while time.time() < deadline:
            m = httpx.get(f"{url}/admin/status", timeout=5.0).json()["metrics"]
            if m["processed_lines"] and m["processed_lines"] >= 1:
                break
            time.sleep(0.1)
        assert m["read_lines"] >= 1
        assert m["processed_lines"] >= 1
        assert m["processing_errors"] == 0
    finally:
        client.close()


def test_service_over_websocket_engine(tmp_path, free_port):
    """A Service listening on ws:// (real RFC6455 framing end-to-end)."""
    settings = ServiceSettings(
        component_type="core",
        engine_addr="ws://127.0.0.1:0",
        http_enabled=False,
        log_dir=tmp_path / "logs",
    )
    service = Service(settings)
    port = service.engine._pair_sock.bound_port
    t = threading.Thread(target=service.run, daemon=True)
    t.start()
    try:
        time.sleep(0.2)
        client = PairDialer(f"ws://127.0.0.1:{port}")
        try:
            assert client.wait_connected(5.0)
            client.send(b"ws-frame")
            assert client.recv(timeout_ms=3000) == b"ws-frame"
        finally:
            client.close()
    finally:
        service.shutdown()
        t.join(timeout=5.0)
