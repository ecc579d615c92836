"""Property-based tests (hypothesis): codec roundtrips, matcher soundness,
C++/Python codec agreement on arbitrary inputs."""
import math

import pytest
from hypothesis import given, settings, strategies as st

from detectmateservice_amd import ops
from detectmateservice_amd.library.parsers.template_matcher import (
    match_template,
    split_template,
)
from detectmateservice_amd.schemas import DetectorSchema, LogSchema, ParserSchema

_text = st.text(
    alphabet=st.characters(blacklist_categories=("Cs",)), max_size=200
)
_i32 = st.integers(min_value=-(2**31), max_value=2**31 - 1)


@settings(max_examples=200, deadline=None)
@given(
    log_id=_text, log=_text, source=_text, host=_text
)
def test_log_schema_roundtrip_property(log_id, log, source, host):
    s = LogSchema(logID=log_id, log=log, logSource=source, hostname=host)
    back = LogSchema.deserialize(s.serialize())
    assert back.logID == log_id
    assert back.log == log
    assert back.logSource == source
    assert back.hostname == host


@settings(max_examples=200, deadline=None)
@given(
    event=_i32,
    variables=st.lists(_text, max_size=8),
    fmt_vars=st.dictionaries(
        st.text(min_size=1, max_size=20), _text, max_size=6
    ),
    ts=_i32,
)
def test_parser_schema_roundtrip_property(event, variables, fmt_vars, ts):
    s = ParserSchema(
        EventID=event, variables=variables, logFormatVariables=fmt_vars,
        receivedTimestamp=ts,
    )
    back = ParserSchema.deserialize(s.serialize())
    assert back.EventID == event
    assert back.variables == variables
    assert back.logFormatVariables == fmt_vars
    assert back.receivedTimestamp == ts


@settings(max_examples=100, deadline=None)
@given(
    score=st.floats(allow_nan=False, allow_infinity=False, width=32),
    stamps=st.lists(_i32, max_size=6),
    alerts=st.dictionaries(st.text(min_size=1, max_size=16), _text, max_size=4),
)
def test_detector_schema_roundtrip_property(score, stamps, alerts):
    s = DetectorSchema(score=score, extractedTimestamps=stamps, alertsObtain=alerts)
    back = DetectorSchema.deserialize(s.serialize())
    assert math.isclose(back.score, score, rel_tol=1e-6, abs_tol=1e-30) or (
        score == 0.0 and back.score == 0.0
    )
    assert back.extractedTimestamps == stamps
    assert back.alertsObtain == alerts


@settings(max_examples=200, deadline=None)
@given(
    segments=st.lists(
        st.text(alphabet="abcXY=_ ", min_size=1, max_size=6),
        min_size=1, max_size=5,
    ),
    fillers=st.lists(st.text(alphabet="0123456789/", max_size=8), max_size=6),
)
def test_match_template_soundness(segments, fillers):
    """If match_template returns captures, substituting them back into the
    template reconstructs the content exactly. (Non-empty segments only:
    adjacent wildcards — "<*><*>" — are inherently ambiguous and collapse,
    in both the Python matcher and the HIP kernel.)"""
    template = "<*>".join(segments)
    segs = split_template(template)
    n_wild = len(segs) - 1
    fill = (fillers + [""] * n_wild)[:n_wild]
    content = segs[0] + "".join(f + s for f, s in zip(fill, segs[1:]))
    caps = match_template(content, segs)
    if caps is None:
        return  # matcher may reject ambiguous constructions; soundness only
    rebuilt = segs[0]
    ci = 0
    for s_ in segs[1:]:
        rebuilt += (caps[ci] if ci < len(caps) else "") + s_
        ci += 1
    assert rebuilt == content


@settings(max_examples=100, deadline=None)
@given(
    lines=st.lists(
        st.binary(max_size=100).map(lambda b: b.replace(b"\x00", b"x")),
        min_size=1, max_size=10,
    )
)
def test_cpp_decode_log_batch_matches_python(lines):
    if not ops.have_extension():
        pytest.skip("extension not built")
    from detectmateservice_amd.ops import _dmx_C

    frames = [
        LogSchema(
            logID=f"id{i}", log=l.decode("latin-1"), logSource="s"
        ).serialize()
        for i, l in enumerate(lines)
    ]
    blines, lens, ids, _s, _h = _dmx_C.decode_log_batch(frames, 128)
    for i, l in enumerate(lines):
        expect = l.decode("latin-1").encode("utf-8")[:128]
        got = bytes(blines[i, : int(lens[i])].numpy().tobytes())
        assert got == expect
        assert bytes(ids[i]) == f"id{i}".encode()


@settings(max_examples=100, deadline=None)
@given(data=st.binary(max_size=300))
def test_decoder_never_crashes_on_garbage(data):
    """Arbitrary bytes must never raise unhandled errors beyond ValueError."""
    for cls in (LogSchema, ParserSchema, DetectorSchema):
        try:
            cls.deserialize(data)
        except ValueError:
            pass


@settings(max_examples=100, deadline=None)
@given(
    payloads=st.lists(st.binary(max_size=300), min_size=1, max_size=8),
    masked=st.booleans(),
)
def test_ws_frame_roundtrip_property(payloads, masked):
    """RFC6455 encode -> _WsFrameReader parse over a real socketpair."""
    import socket as s_mod

    from detectmateservice_amd.engine.sockets import _ws_encode, _WsFrameReader

    a, b = s_mod.socketpair()
    try:
        blob = b"".join(_ws_encode(p, mask=masked) for p in payloads)
        a.sendall(blob)
        a.shutdown(s_mod.SHUT_WR)
        reader = _WsFrameReader(b, server_side=masked)
        got = []
        while True:
            frames = reader.next_frames()
            if frames is None:
                break
            got.extend(frames)
            if len(got) >= len(payloads):
                break
        assert got == payloads
    finally:
        a.close()
        b.close()


@settings(max_examples=100, deadline=None)
@given(payloads=st.lists(st.binary(min_size=0, max_size=100000), min_size=1, max_size=3))
def test_sp_frame_roundtrip_property(payloads):
    """SP 64-bit framing roundtrip through _FrameReader."""
    import socket as s_mod
    import struct as struct_mod

    from detectmateservice_amd.engine.sockets import _FrameReader

    a, b = s_mod.socketpair()
    try:
        blob = b"".join(struct_mod.pack(">Q", len(p)) + p for p in payloads)
        a.sendall(blob)
        a.shutdown(s_mod.SHUT_WR)
        reader = _FrameReader(b, sp=True)
        got = []
        while len(got) < len(payloads):
            frames = reader.next_frames()
            assert frames is not None
            got.extend(frames)
        assert got == payloads
    finally:
        a.close()
        b.close()


@settings(max_examples=50, deadline=None)
@given(
    payloads=st.lists(st.binary(min_size=0, max_size=2000), min_size=1,
                      max_size=20),
    sp=st.booleans(),
)
def test_pack_frames_roundtrip_through_reader(payloads, sp):
    """C++ pack_frames blob parsed back by the C++ FdFrameReader."""
    import socket as s_mod

    from detectmateservice_amd import ops

    if not ops.have_extension():
        pytest.skip("extension not built")
    from detectmateservice_amd.ops import _dmx_C

    a, b = s_mod.socketpair()
    try:
        a.sendall(_dmx_C.pack_frames(payloads, sp))
        a.shutdown(s_mod.SHUT_WR)
        reader = _dmx_C.FdFrameReader(b.fileno(), sp)
        got = []
        while len(got) < len(payloads):
            frames = reader.read_batch(4096, 2000)
            if not frames:
                break
            got.extend(frames)
        assert got == payloads
    finally:
        a.close()
        b.close()


@settings(max_examples=50, deadline=None)
@given(
    payloads=st.lists(st.binary(min_size=0, max_size=5000), min_size=1,
                      max_size=30),
)
def test_shm_ring_roundtrip_property(payloads):
    """Arbitrary frames (incl. empty and multi-KB) through a small ring
    with forced wrap-around."""
    import tempfile as tf

    from detectmateservice_amd import ops

    if not ops.have_extension():
        pytest.skip("extension not built")
    from detectmateservice_amd.ops import _dmx_C

    path = tf.mktemp(prefix="dmx-ring-prop-")
    prod = _dmx_C.ShmRing(path, 16384, True)  # tiny: wraps constantly
    cons = _dmx_C.ShmRing(path, 16384, False)
    try:
        got = []
        sent = 0
        while sent < len(payloads):
            n = prod.write_frames(payloads[sent:])
            sent += n
            while True:
                batch = cons.read_batch(4096, 10)
                if not batch:
                    break
                got.extend(batch)
            if n == 0:
                # frame larger than the ring: skip it (documented drop)
                if len(payloads[sent]) + 4 > 16384:
                    got.append(payloads[sent])
                    sent += 1
                    continue
        while len(got) < len(payloads):
            batch = cons.read_batch(4096, 100)
            if not batch:
                break
            got.extend(batch)
        assert got == payloads
    finally:
        import os as os_mod

        os_mod.unlink(path)
