"""Settings tests (reference shapes: test_component_id.py,
test_config_reading.py, test_tls_settings.py)."""
import pytest

from detectmateservice_amd.settings import ServiceSettings, TlsInputConfig


def test_component_id_stable_from_name():
    a = ServiceSettings(component_type="detector", component_name="d1")
    b = ServiceSettings(component_type="detector", component_name="d1")
    assert a.component_id == b.component_id
    c = ServiceSettings(component_type="detector", component_name="d2")
    assert a.component_id != c.component_id


def test_component_id_from_engine_addr():
    a = ServiceSettings(component_type="parser", engine_addr="tcp://127.0.0.1:5555")
    b = ServiceSettings(component_type="parser", engine_addr="tcp://127.0.0.1:5556")
    assert a.component_id != b.component_id


def test_explicit_component_id_kept():
    s = ServiceSettings(component_id="my-id")
    assert s.component_id == "my-id"


def test_bad_scheme_rejected():
    with pytest.raises(ValueError):
        ServiceSettings(engine_addr="http://127.0.0.1:80")
    with pytest.raises(ValueError):
        ServiceSettings(out_addr=["ftp://x:1"])


def test_tcp_requires_port():
    with pytest.raises(ValueError):
        ServiceSettings(engine_addr="tcp://127.0.0.1")


def test_tls_requires_config():
    with pytest.raises(ValueError):
        ServiceSettings(engine_addr="tls+tcp://127.0.0.1:5555")
    with pytest.raises(ValueError):
        ServiceSettings(out_addr=["tls+tcp://127.0.0.1:5555"])
    # with config it validates
    s = ServiceSettings(
        engine_addr="tls+tcp://127.0.0.1:5555",
        tls_input=TlsInputConfig(cert_key_file="/tmp/cert.pem"),
    )
    assert s.tls_input is not None


def test_extra_fields_forbidden():
    with pytest.raises(Exception):
        ServiceSettings(bogus_field=1)


def test_from_yaml_and_env_precedence(tmp_path, monkeypatch):
    """Env vars win over YAML (reference settings.py:151-168)."""
    cfg = tmp_path / "settings.yaml"
    cfg.write_text(
        "component_type: parser\n"
        "component_name: from-yaml\n"
        "engine_recv_timeout: 200\n"
    )
    monkeypatch.setenv("DETECTMATE_COMPONENT_NAME", "from-env")
    monkeypatch.setenv("DETECTMATE_ENGINE_RETRY_COUNT", "7")
    s = ServiceSettings.from_yaml(cfg)
    assert s.component_name == "from-env"
    assert s.engine_recv_timeout == 200
    assert s.engine_retry_count == 7
    assert isinstance(s.engine_retry_count, int)


def test_nested_env_override(tmp_path, monkeypatch):
    cfg = tmp_path / "settings.yaml"
    cfg.write_text("engine_addr: tls+tcp://127.0.0.1:5555\n")
    monkeypatch.setenv("DETECTMATE_TLS_INPUT__CERT_KEY_FILE", "/tmp/c.pem")
    s = ServiceSettings.from_yaml(cfg)
    assert str(s.tls_input.cert_key_file) == "/tmp/c.pem"


def test_out_addr_env_list(monkeypatch):
    monkeypatch.setenv(
        "DETECTMATE_OUT_ADDR", "ipc:///tmp/a.ipc, tcp://127.0.0.1:7000"
    )
    s = ServiceSettings.from_env()
    assert s.out_addr == ["ipc:///tmp/a.ipc", "tcp://127.0.0.1:7000"]


def test_bounds():
    with pytest.raises(Exception):
        ServiceSettings(engine_retry_count=0)
    with pytest.raises(Exception):
        ServiceSettings(engine_buffer_size=9000)


def test_config_class_override(tmp_path, monkeypatch):
    """settings.config_class overrides the resolver's convention-based
    config class (reference settings.py:52)."""
    from detectmateservice_amd import Service

    svc = Service(ServiceSettings(
        component_type="NewValueDetector",
        config_class="detectors.new_value.NewValueDetectorConfig",
        engine_addr=f"ipc://{tmp_path}/cc.ipc",
        http_enabled=False,
        log_dir=tmp_path / "logs",
    ))
    try:
        assert svc.config_manager.schema is not None
        assert svc.config_manager.schema.__name__ == "NewValueDetectorConfig"
    finally:
        svc.engine.close()


def test_shm_scheme_accepted_tls_rejected():
    s = ServiceSettings(engine_addr="shm:///dmx-x", out_addr=["shm:///dmx-y"])
    assert s.engine_addr == "shm:///dmx-x"
    # shm listener refuses TLS config at socket construction
    from detectmateservice_amd.engine.sockets import ShmListener

    with pytest.raises(ValueError):
        ShmListener("shm:///dmx-x", tls_config=object())
