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
