"""Checkpoint/resume tests: Service-level save/restore of detector state
over the admin API and directly (SURVEY.md §5.4 — new capability)."""
import threading

import httpx
import pytest
import yaml

from detectmateservice_amd import Service, ServiceSettings
from detectmateservice_amd.schemas import ParserSchema


def _detector_settings(tmp_path, ipc_addr, port=0, http=False, ck_dir=None):
    cfg_file = tmp_path / "det.yaml"
    cfg_file.write_text(yaml.safe_dump({
        "detectors": {"NewValueDetector": {
            "method_type": "new_value_detector",
            "data_use_training": 2,
            "global": {"g": {"header_variables": [{"pos": "URL"}]}},
        }}
    }))
    return ServiceSettings(
        component_type="NewValueDetector",
        engine_addr=ipc_addr,
        config_file=cfg_file,
        http_enabled=http,
        http_port=port,
        log_dir=tmp_path / "logs",
        checkpoint_dir=ck_dir,
    )


def _frame(url, lid="x"):
    return ParserSchema(EventID=1, logID=lid, logFormatVariables={"URL": url}).serialize()


def test_service_checkpoint_restore_roundtrip(tmp_path, ipc_addr):
    svc = Service(_detector_settings(tmp_path, ipc_addr))
    try:
        # train through the service processor
        svc.process_batch([_frame("/a"), _frame("/b")])
        assert svc.process(_frame("/a")) is None
        ck = tmp_path / "ckpt" / "det.pt"
        info = svc.checkpoint(ck)
        assert ck.exists()

        # new service instance: restores learned state
        svc2 = Service(_detector_settings(tmp_path, ipc_addr + "-2"))
        try:
            # untrained service would still be in training phase; restore
            # brings seen_lines and the known set back
            svc2.restore(ck)
            assert svc2.process(_frame("/a")) is None       # known
            assert svc2.process(_frame("/evil")) is not None  # alert
        finally:
            svc2.engine.close()
    finally:
        svc.engine.close()


def test_restore_rejects_wrong_component_type(tmp_path, ipc_addr):
    svc = Service(_detector_settings(tmp_path, ipc_addr))
    try:
        ck = tmp_path / "det.pt"
        svc.checkpoint(ck)
        other = Service(ServiceSettings(
            component_type="core", engine_addr=ipc_addr + "-3",
            http_enabled=False, log_dir=tmp_path / "logs",
        ))
        try:
            with pytest.raises(ValueError):
                other.restore(ck)
        finally:
            other.engine.close()
    finally:
        svc.engine.close()


def test_checkpoint_over_http(tmp_path, ipc_addr, free_port):
    """Admin checkpoint/restore operate on names INSIDE the configured
    checkpoint_dir; traversal and absolute escapes are refused with 403."""
    ck_dir = tmp_path / "ckpts"
    settings = _detector_settings(
        tmp_path, ipc_addr, port=free_port, http=True, ck_dir=ck_dir)
    svc = Service(settings)
    t = threading.Thread(target=svc.run, daemon=True)
    t.start()
    try:
        assert svc.web_server.wait_started(10.0)
        svc.process_batch([_frame("/a"), _frame("/b")])
        url = f"http://127.0.0.1:{free_port}"
        r = httpx.post(f"{url}/admin/checkpoint", json={"path": "http_ck.pt"},
                       timeout=10.0)
        assert r.status_code == 200
        assert (ck_dir / "http_ck.pt").exists()
        r = httpx.post(f"{url}/admin/restore", json={"path": "http_ck.pt"},
                       timeout=10.0)
        assert r.status_code == 200
        # traversal out of checkpoint_dir: refused
        r = httpx.post(f"{url}/admin/restore",
                       json={"path": "../det.yaml"}, timeout=10.0)
        assert r.status_code == 403
        # absolute path outside checkpoint_dir: refused
        r = httpx.post(f"{url}/admin/checkpoint",
                       json={"path": "/etc/cron.d/x.pt"}, timeout=10.0)
        assert r.status_code == 403
    finally:
        svc.shutdown()
        t.join(timeout=5.0)


def test_checkpoint_over_http_refused_without_dir(tmp_path, ipc_addr, free_port):
    """No checkpoint_dir configured => the unauthenticated admin surface
    must not touch the filesystem at all."""
    settings = _detector_settings(tmp_path, ipc_addr, port=free_port, http=True)
    svc = Service(settings)
    t = threading.Thread(target=svc.run, daemon=True)
    t.start()
    try:
        assert svc.web_server.wait_started(10.0)
        url = f"http://127.0.0.1:{free_port}"
        for ep in ("checkpoint", "restore"):
            r = httpx.post(f"{url}/admin/{ep}", json={"path": "x.pt"},
                           timeout=10.0)
            assert r.status_code == 403
    finally:
        svc.shutdown()
        t.join(timeout=5.0)


def test_restore_refuses_pickle_payload(tmp_path, ipc_addr):
    """A checkpoint file carrying an arbitrary pickle object (the RCE
    vector ADVICE flagged) must raise on load, not execute."""
    import torch

    class Evil:  # torch.save will pickle this; weights_only load must balk
        def __reduce__(self):
            return (print, ("pwned",))

    ck = tmp_path / "evil.pt"
    torch.save({"component_type": "NewValueDetector", "component_state": Evil()}, ck)
    svc = Service(_detector_settings(tmp_path, ipc_addr))
    try:
        with pytest.raises(Exception):
            svc.restore(ck)
    finally:
        svc.engine.close()


def test_reconfigure_reload_rebuilds_component(tmp_path):
    """reconfigure(reload=True) swaps in a component built from the NEW
    config (the default reference-parity behavior keeps the old one)."""
    from detectmateservice_amd import Service, ServiceSettings

    cfg_file = tmp_path / "cfg.yaml"
    cfg_file.write_text(
        "detectors:\n  RandomDetector:\n    params:\n      threshold: 0.5\n"
    )
    svc = Service(ServiceSettings(
        component_type="RandomDetector",
        engine_addr=f"ipc://{tmp_path}/rl.ipc",
        http_enabled=False,
        config_file=cfg_file,
        log_dir=tmp_path / "logs",
    ))
    try:
        first = svc.library_component
        # plain reconfigure: live component unchanged (reference parity)
        svc.reconfigure({"detectors": {"RandomDetector": {"params": {"threshold": 0.9}}}})
        assert svc.library_component is first
        # reload: a NEW component instance built from the new config
        svc.reconfigure(
            {"detectors": {"RandomDetector": {"params": {"threshold": 0.9}}}},
            reload=True,
        )
        assert svc.library_component is not first
        assert svc.library_component.config.params["threshold"] == 0.9
    finally:
        svc.engine.close()
