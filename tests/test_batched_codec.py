"""Batched C++ codec + batched MatcherParser path vs the per-message
Python path (both must produce semantically identical ParserSchema)."""
import pytest
import torch

from detectmateservice_amd import ops
from detectmateservice_amd.library.parsers import MatcherParser
from detectmateservice_amd.schemas import LogSchema, ParserSchema
from detectmateservice_amd.utils.synthetic import (
    AUDIT_LOG_FORMAT,
    AUDIT_TEMPLATES,
    AuditLogGenerator,
)

needs_ext = pytest.mark.skipif(
    not ops.have_extension(), reason="C++ extension not built"
)


@needs_ext
def test_decode_log_batch():
    from detectmateservice_amd.ops import _dmx_C

    frames = [
        LogSchema(logID=f"id{i}", log=f"line number {i}", logSource="s",
                  hostname="h").serialize()
        for i in range(5)
    ]
    lines, lens, ids, sources, hosts = _dmx_C.decode_log_batch(frames, 64)
    assert lines.shape == (5, 64)
    for i in range(5):
        text = bytes(lines[i, : int(lens[i])].numpy().tobytes()).decode()
        assert text == f"line number {i}"
        assert bytes(ids[i]) == f"id{i}".encode()


@needs_ext
def test_batched_parser_equals_python_parser():
    gen = AuditLogGenerator(seed=31)
    frames = [
        LogSchema(logID=f"L{i}", log=gen.line()[0]).serialize() for i in range(64)
    ]
    parser = MatcherParser(
        {"log_format": AUDIT_LOG_FORMAT, "templates": list(AUDIT_TEMPLATES)}
    )
    batched = parser.process_batch(frames)      # C++ codec path (>=8 frames)
    python = parser._process_python(frames)

    for i, (b, p) in enumerate(zip(batched, python)):
        pb = ParserSchema.deserialize(b)
        pp = ParserSchema.deserialize(p)
        assert pb.EventID == pp.EventID, i
        assert pb.variables == pp.variables, i
        assert pb.logID == pp.logID
        assert pb.log == pp.log
        assert pb.template == pp.template
        assert pb.logFormatVariables == pp.logFormatVariables, i
        assert pb.parserType == pp.parserType


@needs_ext
def test_batched_parser_negative_event_id():
    parser = MatcherParser({"templates": ["nomatch=<*>"]})
    frames = [LogSchema(logID=str(i), log="zzz unmatched").serialize() for i in range(10)]
    out = parser.process_batch(frames)
    for b in out:
        pb = ParserSchema.deserialize(b)
        assert pb.EventID == -1
        assert pb.template == ""
        assert pb.variables == []


def test_transformer_detector_cpu():
    from detectmateservice_amd.library.detectors import TransformerDetector

    det = TransformerDetector(
        {"data_use_training": 32, "z_threshold": 4.0, "device": "cpu"}
    )
    gen = AuditLogGenerator(seed=41)

    def frame(line, lid):
        return ParserSchema(logID=lid, log=line, EventID=1).serialize()

    train = [frame(gen.line()[0], f"t{i}") for i in range(32)]
    out = det.process_batch(train)
    assert all(o is None for o in out)
    assert det.threshold is not None

    normal = [frame(gen.line()[0], f"n{i}") for i in range(16)]
    out = det.process_batch(normal)
    # calibrated at 4 sigma: normal traffic should mostly pass
    assert sum(o is not None for o in out) <= 2

    # wildly different line content should push the score distribution
    state = det.state_dict()
    det2 = TransformerDetector(
        {"data_use_training": 0, "z_threshold": 4.0, "device": "cpu"}
    )
    det2.load_state_dict(state)
    out2 = det2.process_batch(normal)
    assert [o is None for o in out] == [o is None for o in out2]


def test_transformer_detector_resolvable():
    from detectmateservice_amd.components.resolver import ComponentResolver

    path, cfg = ComponentResolver().resolve("TransformerDetector")
    assert path.endswith("TransformerDetector")
    assert cfg.endswith("TransformerDetectorConfig")


def test_embedding_detector_cpu():
    from detectmateservice_amd.library.detectors import EmbeddingDetector

    det = EmbeddingDetector({"data_use_training": 48, "z_threshold": 4.0,
                             "device": "cpu"})
    gen = AuditLogGenerator(seed=61)

    def frame(line, lid):
        return ParserSchema(logID=lid, log=line, EventID=1).serialize()

    train = [frame(gen.line()[0], f"t{i}") for i in range(48)]
    assert all(o is None for o in det.process_batch(train))
    assert det._n == 48

    normal = [frame(gen.line()[0], f"n{i}") for i in range(16)]
    out = det.process_batch(normal)
    assert sum(o is not None for o in out) <= 2  # 4-sigma on in-dist data

    # a wildly out-of-distribution line
    weird = frame("\x01\x02!!!" + "Z" * 180, "weird")
    out = det.process_batch(normal[:3] + [weird])
    # distance ordering: the weird line must have the max distance
    emb = det.embed([b"normal audit line", b"\x01\x02!!!" + b"Z" * 180])
    d = det._distance(emb)
    assert d[1] > d[0]

    # checkpoint roundtrip: restored detector scores `normal` identically
    ref = det.process_batch(normal)
    state = det.state_dict()
    det2 = EmbeddingDetector({"data_use_training": 0, "device": "cpu"})
    det2.load_state_dict(state)
    out2 = det2.process_batch(normal)
    assert [o is None for o in ref] == [o is None for o in out2]


def test_embedding_detector_resolvable():
    from detectmateservice_amd.components.resolver import ComponentResolver

    path, cfg = ComponentResolver().resolve("EmbeddingDetector")
    assert path.endswith("EmbeddingDetector")


def test_fused_pipeline_detector_component():
    """LogSchema in -> DetectorSchema alerts out through the fused
    pipeline as ONE component (CPU fallback path here; GPU in -m gpu)."""
    from detectmateservice_amd.library.detectors import FusedPipelineDetector
    from detectmateservice_amd.schemas import DetectorSchema, LogSchema
    from detectmateservice_amd.utils.synthetic import (
        AUDIT_LOG_FORMAT,
        AUDIT_TEMPLATES,
        AuditLogGenerator,
    )

    det = FusedPipelineDetector({
        "templates": list(AUDIT_TEMPLATES),
        "log_format": AUDIT_LOG_FORMAT,
        "watches": [{"kind": "variable", "pos": 5, "event": 1}],
        "data_use_training": 64,
        "use_transformer": False,
        "device": "cpu",
    })
    gen = AuditLogGenerator(seed=5)
    train = [LogSchema(logID=f"t{i}", log=gen.line()[0]).serialize() for i in range(64)]
    assert all(o is None for o in det.process_batch(train))

    normal = [LogSchema(logID=f"n{i}", log=gen.line()[0]).serialize() for i in range(16)]
    assert sum(o is not None for o in det.process_batch(normal)) == 0

    bad = LogSchema(logID="bad", log=(
        "type=USER_ACCT msg=audit(1642723741.072:999): pid=1 uid=0 auid=1 ses=1 "
        "msg='op=PAM:accounting acct=\"intruder\" exe=/usr/sbin/cron hostname=? "
        "addr=? terminal=cron res=success'"
    )).serialize()
    out = det.process_batch(normal[:3] + [bad])
    assert [o is not None for o in out] == [False, False, False, True]
    alert = DetectorSchema.deserialize(out[3])
    assert alert.logIDs == ["bad"]
    assert "unknown watched value" in alert.description

    # checkpoint roundtrip
    state = det.state_dict()
    det2 = FusedPipelineDetector({
        "templates": list(AUDIT_TEMPLATES),
        "log_format": AUDIT_LOG_FORMAT,
        "watches": [{"kind": "variable", "pos": 5, "event": 1}],
        "data_use_training": 0,
        "use_transformer": False,
        "device": "cpu",
    })
    det2.load_state_dict(state)
    out2 = det2.process_batch(normal[:3] + [bad])
    assert [o is not None for o in out2] == [False, False, False, True]


def test_fused_pipeline_detector_resolvable():
    from detectmateservice_amd.components.resolver import ComponentResolver

    path, _ = ComponentResolver().resolve("FusedPipelineDetector")
    assert path.endswith("FusedPipelineDetector")


@needs_ext
def test_decode_log_batch_packed_matches_list_variant():
    from detectmateservice_amd.ops import _dmx_C

    frames = [
        LogSchema(logID=f"id{i}", log=f"the line {i}", logSource="s").serialize()
        for i in range(17)
    ] + [LogSchema(logID="", log="no id").serialize()]
    l1, n1, ids, _s, _h = _dmx_C.decode_log_batch(frames, 64)
    l2, n2, blob, off = _dmx_C.decode_log_batch_packed(frames, 64, False)
    assert torch.equal(l1, l2) and torch.equal(n1, n2)
    for i in range(len(frames)):
        assert blob[int(off[i]):int(off[i + 1])] == bytes(ids[i])


def _varint(v: int) -> bytes:
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


@pytest.mark.skipif(not ops.have_extension(), reason="extension not built")
def test_packed_decode_rejects_overflowing_varint_lengths():
    """Crafted 10-byte varint string lengths near 2^64 must not wrap the
    bounds check (ADVICE high: heap over-read). The decoder stops at the
    malformed field; adjacent valid frames still decode."""
    from detectmateservice_amd.ops import _dmx_C

    evil_frames = []
    for length in (2**64 - 8, 2**64 - 1, 2**63, 2**32):
        # field 3 (log), wire type 2, huge length, tiny payload
        evil_frames.append(bytes([0x1A]) + _varint(length) + b"AB")
        # field 2 (logID) with huge length — span fields
        evil_frames.append(bytes([0x12]) + _varint(length) + b"CD")
    good = LogSchema(logID="ok", log="fine line").serialize()
    frames = evil_frames + [good]
    lines, lens, blob, off = _dmx_C.decode_log_batch_packed(frames, 64, False)
    assert lines.shape == (len(frames), 64)
    # malformed frames contribute empty/zero rows, never OOB copies
    for i in range(len(evil_frames)):
        assert int(lens[i]) == 0
        assert blob[int(off[i]):int(off[i + 1])] == b""
    assert bytes(lines[-1, : int(lens[-1])].numpy().tobytes()) == b"fine line"
    assert blob[int(off[len(frames) - 1]):int(off[len(frames)])] == b"ok"


@pytest.mark.skipif(not ops.have_extension(), reason="extension not built")
def test_packed_decode_rejects_overflowing_skip_lengths():
    """skip_field wire-type 2 with a wrapping length (unknown field) must
    terminate the walk instead of advancing pos past the buffer."""
    from detectmateservice_amd.ops import _dmx_C

    # unknown field 9, wt 2, huge length, then a valid log field that the
    # decoder must NOT reach (the frame is malformed from field 9 on)
    f = bytes([0x4A]) + _varint(2**64 - 2) + bytes([0x1A, 0x03]) + b"abc"
    lines, lens, blob, off = _dmx_C.decode_log_batch_packed([f], 64, False)
    assert int(lens[0]) == 0

    # fixed-width skips at the buffer edge: wt 1 with <8 bytes left,
    # wt 5 with <4 bytes left
    for frame in (bytes([0x49]) + b"\x01\x02", bytes([0x4D]) + b"\x01"):
        lines, lens, blob, off = _dmx_C.decode_log_batch_packed([frame], 64, False)
        assert int(lens[0]) == 0


@pytest.mark.skipif(not ops.have_extension(), reason="extension not built")
def test_parser_hash_decode_rejects_overflowing_varints():
    """The ParserSchema watch-hash decoder shares the same varint walk;
    crafted lengths in variables / map entries must not over-read."""
    from detectmateservice_amd.ops import _dmx_C

    evil = bytes([0x32]) + _varint(2**64 - 4) + b"XY"  # field 6 repeated var
    good = ParserSchema(EventID=1, logID="g", variables=["v0"]).serialize()
    hashes, event_ids, log_ids = _dmx_C.parser_watch_hashes(
        [evil, good], [(-1, 0)], [], False)
    assert int(event_ids[1]) == 1
