"""Detector tests: NewValue train/detect semantics, combo, dummy
alternation, random determinism, checkpoint/resume."""
import pytest

from detectmateservice_amd.library.detectors import (
    DummyDetector,
    NewValueComboDetector,
    NewValueDetector,
    RandomDetector,
)
from detectmateservice_amd.schemas import DetectorSchema, ParserSchema


def _parsed(url="/index.html", event_id=1, variables=None, log_id="l1"):
    return ParserSchema(
        EventID=event_id,
        variables=variables or [],
        logID=log_id,
        logFormatVariables={"URL": url},
    ).serialize()


NV_CONFIG = {
    "data_use_training": 2,
    "global": {"global_instance": {"header_variables": [{"pos": "URL"}]}},
}


def test_new_value_detector_train_then_detect():
    det = NewValueDetector(NV_CONFIG)
    # first two frames are training (data_use_training: 2)
    out = det.process_batch([_parsed("/a"), _parsed("/b")])
    assert out == [None, None]
    # known value: no alert
    assert det.process(_parsed("/a")) is None
    # unseen value: alert with the reference's description shape
    alert_bytes = det.process(_parsed("/foobar", log_id="L9"))
    assert alert_bytes is not None
    alert = DetectorSchema.deserialize(alert_bytes)
    assert alert.description == "Unknown value: '/foobar'"
    assert "Global - URL" in alert.alertsObtain
    assert alert.logIDs == ["L9"]
    assert alert.score == 1.0


def test_new_value_training_spans_batches():
    det = NewValueDetector(NV_CONFIG)
    assert det.process(_parsed("/a")) is None  # train 1
    assert det.process(_parsed("/b")) is None  # train 2
    assert det.process(_parsed("/c")) is not None  # detect: unseen


def test_new_value_event_scoped_variables():
    cfg = {
        "data_use_training": 1,
        "events": {
            1: {"inst": {"variables": [{"pos": 0, "name": "var1"}]}}
        },
    }
    det = NewValueDetector(cfg)
    assert det.process(_parsed(variables=["ok"], event_id=1)) is None  # train
    assert det.process(_parsed(variables=["ok"], event_id=1)) is None
    assert det.process(_parsed(variables=["bad"], event_id=1)) is not None
    # other events are not watched
    assert det.process(_parsed(variables=["bad"], event_id=2)) is None


def test_new_value_checkpoint_resume():
    det = NewValueDetector(NV_CONFIG)
    det.process_batch([_parsed("/a"), _parsed("/b")])
    state = det.state_dict()

    det2 = NewValueDetector(NV_CONFIG)
    det2.load_state_dict(state)
    assert det2.process(_parsed("/a")) is None
    assert det2.process(_parsed("/zzz")) is not None


def test_new_value_combo_detector():
    cfg = {
        "data_use_training": 1,
        "global": {"g": {"header_variables": [{"pos": "URL"}],
                          "variables": [{"pos": 0, "name": "v0"}]}},
    }
    det = NewValueComboDetector(cfg)
    assert det.process(_parsed("/a", variables=["x"])) is None  # train
    assert det.process(_parsed("/a", variables=["x"])) is None  # known combo
    # known parts, new combination
    out = det.process(_parsed("/a", variables=["y"]))
    assert out is not None


def test_dummy_detector_alternates():
    det = DummyDetector()
    results = [det.process(_parsed()) is not None for _ in range(4)]
    assert results == [False, True, False, True]
    alert = DetectorSchema.deserialize(det.process_batch([_parsed()] * 2)[1])
    assert alert.description == "Dummy detection process"
    assert alert.score == 1.0


def test_random_detector_seeded():
    cfg = {"params": {"seed": 42, "threshold": 0.5}}
    a = RandomDetector(cfg)
    b = RandomDetector(cfg)
    frames = [_parsed(log_id=str(i)) for i in range(50)]
    ra = [x is not None for x in a.process_batch(frames)]
    rb = [x is not None for x in b.process_batch(frames)]
    assert ra == rb
    assert any(ra) and not all(ra)


def test_random_detector_threshold_extremes():
    never = RandomDetector({"params": {"seed": 1, "threshold": 1.1}})
    assert all(x is None for x in never.process_batch([_parsed()] * 20))
    always = RandomDetector({"params": {"seed": 1, "threshold": -0.1}})
    assert all(x is not None for x in always.process_batch([_parsed()] * 20))


def test_method_type_mismatch_raises():
    with pytest.raises(ValueError):
        NewValueDetector({"method_type": "wrong_type"})


def test_buffer_modes():
    """FIXED emits once per full window; SLIDING detects per line once
    primed (reference BufferMode, interfaces.md:167)."""
    from detectmateservice_amd.components.base import CoreDetector, CoreDetectorConfig

    class WindowCounter(CoreDetector):
        CONFIG_CLASS = CoreDetectorConfig

        def __init__(self, config=None):
            super().__init__(config)
            self.windows = []

        def train(self, batch):
            pass

        def detect(self, parsed, alert):
            alert.description = "line"
            return True

        def detect_window(self, window, alert):
            self.windows.append([p.logID for p in window])
            alert.description = f"window of {len(window)}"
            return True

    frames = [_parsed(log_id=f"l{i}") for i in range(8)]

    fixed = WindowCounter({"buffer_mode": "fixed", "buffer_size": 4})
    out = fixed.process_batch(frames)
    assert [o is not None for o in out] == [False] * 3 + [True] + [False] * 3 + [True]
    assert fixed.windows == [["l0", "l1", "l2", "l3"], ["l4", "l5", "l6", "l7"]]

    sliding = WindowCounter({"buffer_mode": "sliding", "buffer_size": 4})
    out = sliding.process_batch(frames)
    assert [o is not None for o in out] == [False] * 3 + [True] * 5
    assert sliding.windows[0] == ["l0", "l1", "l2", "l3"]
    assert sliding.windows[-1] == ["l4", "l5", "l6", "l7"]

    nobuf = WindowCounter({})
    out = nobuf.process_batch(frames)
    assert all(o is not None for o in out)
    assert nobuf.windows == []


def test_sklearn_detector_isolation_forest():
    """Train on regular audit lines; a wildly different line is flagged."""
    from detectmateservice_amd.library.detectors import SklearnDetector
    from detectmateservice_amd.schemas import DetectorSchema, ParserSchema
    from detectmateservice_amd.utils.synthetic import AuditLogGenerator

    det = SklearnDetector({
        "data_use_training": 128,
        "params": {"model": "isolation_forest", "contamination": 0.01,
                   "seed": 7},
    })
    gen = AuditLogGenerator(seed=11)

    def frame(line, lid):
        return ParserSchema(logID=lid, log=line, EventID=1).serialize()

    train = [frame(gen.line()[0], f"t{i}") for i in range(128)]
    assert all(o is None for o in det.process_batch(train))

    normal = [frame(gen.line()[0], f"n{i}") for i in range(24)]
    fps = sum(o is not None for o in det.process_batch(normal))
    assert fps <= 4  # contamination-bounded false positives

    weird = frame("\x01\x02" + "Z" * 400 + "!!!$$$%%%", "weird")
    out = det.process_batch([weird])
    assert out[0] is not None
    alert = DetectorSchema.deserialize(out[0])
    assert alert.logIDs == ["weird"]
    assert "isolation_forest" in alert.description

    # checkpoint roundtrip
    state = det.state_dict()
    det2 = SklearnDetector({"data_use_training": 0,
                            "params": {"model": "isolation_forest"}})
    det2.load_state_dict(state)
    assert det2.process_batch([weird])[0] is not None


def test_frequency_detector_flood():
    """A flood of one EventID inside a window raises a rate alert."""
    from detectmateservice_amd.library.detectors import FrequencyDetector
    from detectmateservice_amd.schemas import DetectorSchema, ParserSchema

    det = FrequencyDetector({
        "data_use_training": 0,
        "window_lines": 100,
        "z_threshold": 4.0,
        "min_windows": 3,
    })

    def frames(ev, n, tag):
        return [ParserSchema(logID=f"{tag}{i}", log="x", EventID=ev).serialize()
                for i in range(n)]

    # steady state: ~50/50 split between events 1 and 2, several windows
    alerts = []
    for w in range(6):
        batch = frames(1, 50, f"a{w}-") + frames(2, 50, f"b{w}-")
        alerts += [o for o in det.process_batch(batch) if o is not None]
    assert alerts == []

    # checkpoint the PRE-flood baseline (after a flood the EWMA adapts,
    # which is the intended behavior)
    state = det.state_dict()

    # flood: event 2 takes the whole window
    out = det.process_batch(frames(2, 100, "flood"))
    flood_alerts = [o for o in out if o is not None]
    assert len(flood_alerts) == 1
    alert = DetectorSchema.deserialize(flood_alerts[0])
    assert "Rate anomaly" in alert.description
    assert "event 2" in alert.description

    # restored detector carries the same baseline -> same flood alert
    det2 = FrequencyDetector({"data_use_training": 0, "window_lines": 100,
                              "z_threshold": 4.0, "min_windows": 3})
    det2.load_state_dict(state)
    out2 = det2.process_batch(frames(2, 100, "flood2"))
    assert sum(o is not None for o in out2) == 1


def test_sklearn_frequency_resolvable():
    from detectmateservice_amd.components.resolver import ComponentResolver

    for name in ("SklearnDetector", "FrequencyDetector"):
        path, _ = ComponentResolver().resolve(name)
        assert path.endswith(name)


def test_tokenizer_detector_compression_surprise():
    """TokenizerDetector: lines fitting the learned BPE structure score
    low; structurally novel lines exceed the z-threshold and alert."""
    from detectmateservice_amd.library.detectors.tokenizer_llm import (
        TokenizerDetector,
    )
    from detectmateservice_amd.schemas import DetectorSchema, ParserSchema

    det = TokenizerDetector({
        "method_type": "tokenizer_detector",
        "vocab_size": 512,
        "z_threshold": 3.0,
        "data_use_training": 0,
    })
    train = [
        ParserSchema(logID=f"t{i}",
                     log=f"type=SYSCALL msg=audit({i}.0:1): arch=c000003e "
                         f"syscall=59 success=yes exit=0 pid={i}")
        for i in range(400)
    ]
    det.train(train)
    normal = ParserSchema(
        logID="n", log="type=SYSCALL msg=audit(999.0:1): arch=c000003e "
                       "syscall=59 success=yes exit=0 pid=999")
    a = DetectorSchema()
    assert det.detect(normal, a) is False
    weird = ParserSchema(logID="w", log="\x01\x02zq9!jx@#kk%%&&~~``\x7f" * 6)
    a = DetectorSchema()
    assert det.detect(weird, a) is True
    assert a.detectorType == "tokenizer_detector" and a.score > 3.0

    # checkpoint roundtrip keeps the learned tokenizer + calibration
    st = det.state_dict()
    det2 = TokenizerDetector({"method_type": "tokenizer_detector"})
    det2.load_state_dict(st)
    a = DetectorSchema()
    assert det2.detect(weird, a) is True


def test_llm_assist_detector_enriches_alerts_via_stub():
    """LLMAssistDetector: inner detector fires; the alert is enriched
    through an OpenAI-compatible stub endpoint; LLM failure never
    suppresses the alert."""
    import http.server
    import json as json_mod
    import threading

    from detectmateservice_amd.library.detectors.tokenizer_llm import (
        LLMAssistDetector,
    )
    from detectmateservice_amd.schemas import DetectorSchema, ParserSchema

    class Stub(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            n = int(self.headers["Content-Length"])
            body = json_mod.loads(self.rfile.read(n))
            assert body["messages"][0]["role"] == "user"
            out = json_mod.dumps({"choices": [{"message": {
                "content": "unseen account value"}}]}).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(out)))
            self.end_headers()
            self.wfile.write(out)

        def log_message(self, *a):  # quiet
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), Stub)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        det = LLMAssistDetector({
            "method_type": "llm_assist_detector",
            "inner": "new_value_detector",
            "llm_base_url": f"http://127.0.0.1:{srv.server_port}",
            "data_use_training": 1,
            "global": {"g": {"header_variables": [{"pos": "URL"}]}},
        })
        det.train([ParserSchema(EventID=1, logID="t",
                                logFormatVariables={"URL": "/ok"})])
        hit = ParserSchema(EventID=1, logID="x",
                           logFormatVariables={"URL": "/evil"})
        a = DetectorSchema()
        assert det.detect(hit, a) is True
        assert "LLM: unseen account value" in a.description
        assert a.alertsObtain.get("llm_summary") == "unseen account value"
    finally:
        srv.shutdown()

    # endpoint down: alert still fires, description unenriched
    det2 = LLMAssistDetector({
        "method_type": "llm_assist_detector",
        "inner": "new_value_detector",
        "llm_base_url": "http://127.0.0.1:1",  # refused
        "llm_timeout_s": 0.3,
        "data_use_training": 1,
        "global": {"g": {"header_variables": [{"pos": "URL"}]}},
    })
    det2.train([ParserSchema(EventID=1, logID="t",
                             logFormatVariables={"URL": "/ok"})])
    a = DetectorSchema()
    assert det2.detect(ParserSchema(EventID=1, logID="y",
                                    logFormatVariables={"URL": "/bad"}),
                       a) is True
    assert "LLM:" not in (a.description or "")
