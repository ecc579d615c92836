"""dist_mode + GPU compute end to end: a fanout topology where the sink
rank runs the FUSED GPU pipeline (cuda) while the collective transport
runs gloo (two processes share the box's single GPU — the engine's dist
loops pick the comm device from the backend, independent of the
component's compute device)."""
import multiprocessing as mp
import os
import time

import pytest
import torch

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, tmp, q):
    try:
        os.environ.update({
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "RANK": str(rank), "WORLD_SIZE": str(world),
        })
        result = _body(rank, world, tmp)
        q.put((rank, "ok", result))
    except Exception:  # noqa: BLE001
        import traceback

        q.put((rank, "err", traceback.format_exc()))


def _body(rank, world, tmp):
    import threading

    import yaml

    from detectmateservice_amd import Service, ServiceSettings
    from detectmateservice_amd.engine.sockets import (
        PairDialer, PairListener, RecvTimeout,
    )
    from detectmateservice_amd.schemas import DetectorSchema, LogSchema
    from detectmateservice_amd.utils.synthetic import (
        AUDIT_LOG_FORMAT,
        AUDIT_TEMPLATES,
        AuditLogGenerator,
    )

    common = dict(
        engine_addr=f"ipc://{tmp}/gfan-{{rank}}.ipc",
        http_enabled=False,
        log_dir=f"{tmp}/logs",
        engine_recv_timeout=50,
        engine_batch_linger_ms=5.0,
        dist_backend="gloo",
        dist_mode="fanout",
    )
    if rank == 0:
        settings = ServiceSettings(component_type="core", **common)
        svc = Service(settings)
    else:
        cfg = f"{tmp}/fused-{rank}.yaml"
        with open(cfg, "w") as fh:
            yaml.safe_dump({"detectors": {"FusedPipelineDetector": {
                "method_type": "fused_pipeline_detector",
                "templates": list(AUDIT_TEMPLATES),
                "log_format": AUDIT_LOG_FORMAT,
                "watches": [{"kind": "variable", "pos": 5, "event": 1}],
                "data_use_training": 64,
                "use_transformer": True,
                "score_threshold": 1.0e9,  # NewValue alerts only
                "device": "cuda:0",
            }}}, fh)
        settings = ServiceSettings(
            component_type="FusedPipelineDetector", config_file=cfg,
            out_addr=[f"ipc://{tmp}/gfan-alerts.ipc"], **common)
        svc = Service(settings)
    t = threading.Thread(target=svc.run, daemon=True)
    t.start()
    try:
        gen = AuditLogGenerator(seed=5, anomaly_rate=0.0)
        if rank == 0:
            time.sleep(1.0)
            feeder = PairDialer(svc.settings.engine_addr)
            assert feeder.wait_connected(15.0)
            # train frames, then one guaranteed-novel watched value
            train = [LogSchema(logID=f"t{i}", log=gen.line()[0]).serialize()
                     for i in range(64)]
            assert feeder.send_many(train, block=True) == 64
            time.sleep(2.0)
            bad = ('type=USER_LOGIN msg=audit(1.2:3): pid=9 uid=0 auid=4 '
                   'ses=2 msg=\'op=PAM:session_open acct="zzz_never_seen" '
                   'exe=/usr/bin/zzz hostname=h addr=1.2.3.4 terminal=tty '
                   'res=success\'')
            feeder.send(LogSchema(logID="evil", log=bad).serialize(),
                        block=True)
            time.sleep(4.0)
            feeder.close()
            return "fed"
        # sink: its fused GPU pipeline consumes the broadcast; alerts
        # appear on ITS out_addr
        sink = PairListener(f"ipc://{tmp}/gfan-alerts.ipc")
        alert = None
        deadline = time.monotonic() + 30
        while alert is None and time.monotonic() < deadline:
            try:
                alert = sink.recv(timeout_ms=500)
            except RecvTimeout:
                continue
        sink.close()
        assert alert is not None, "no alert from the GPU sink"
        d = DetectorSchema.deserialize(alert)
        assert d.detectorType == "fused_pipeline_detector"
        assert "evil" in (d.logIDs or [""])[0]
        return "gpu-sink-alerted"
    finally:
        svc.shutdown()
        t.join(timeout=10.0)


def test_fanout_with_gpu_fused_sink(tmp_path):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29541
    procs = [ctx.Process(target=_worker, args=(r, 2, port, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(2):
            rank, status, payload = q.get(timeout=240)
            assert status == "ok", f"rank {rank} failed:\n{payload}"
            results[rank] = payload
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    assert results[0] == "fed"
    assert results[1] == "gpu-sink-alerted"
