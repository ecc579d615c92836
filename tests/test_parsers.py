"""MatcherParser tests: log_format extraction, template matching,
normalization; synthetic audit-log end-to-end."""
from detectmateservice_amd.library.parsers import DummyParser, MatcherParser
from detectmateservice_amd.library.parsers.template_matcher import (
    compile_log_format,
    match_template,
    split_template,
)
from detectmateservice_amd.schemas import LogSchema, ParserSchema
from detectmateservice_amd.utils.synthetic import (
    AUDIT_LOG_FORMAT,
    AUDIT_TEMPLATES,
    AuditLogGenerator,
)


def test_log_format_compile_and_match():
    fmt = '<IP> - - [<Time>] "<Method> <URL> <Protocol>" <Status> <Bytes>'
    rx = compile_log_format(fmt)
    line = '1.2.3.4 - - [10/Oct/2000:13:55:36] "GET /index.html HTTP/1.0" 200 2326'
    m = rx.match(line)
    assert m
    assert m.group("IP") == "1.2.3.4"
    assert m.group("Method") == "GET"
    assert m.group("URL") == "/index.html"
    assert m.group("Status") == "200"
    assert m.group("Bytes") == "2326"


def test_template_match_basic():
    segs = split_template("pid=<*> uid=<*> res=<*>")
    caps = match_template("pid=123 uid=0 res=success", segs)
    assert caps == ["123", "0", "success"]


def test_template_match_anchoring():
    segs = split_template("pid=<*> uid=<*>")
    # must start with "pid=" and end after uid capture
    assert match_template("xpid=1 uid=2", segs) is None
    assert match_template("pid=1 uid=2", segs) == ["1", "2"]


def test_template_no_match():
    segs = split_template("completely=<*> different=<*>")
    assert match_template("pid=1 uid=2", segs) is None


def test_template_trailing_literal():
    segs = split_template("a=<*> end")
    assert match_template("a=1 end", segs) == ["1"]
    assert match_template("a=1 endx", segs) is None


def test_matcher_parser_on_synthetic_audit():
    parser = MatcherParser(
        {
            "log_format": AUDIT_LOG_FORMAT,
            "templates": AUDIT_TEMPLATES,
        }
    )
    gen = AuditLogGenerator(seed=7)
    n_match = 0
    for _ in range(200):
        line, _anom, tidx = gen.line()
        frame = LogSchema(logID="x", log=line).serialize()
        out = parser.process(frame)
        parsed = ParserSchema.deserialize(out)
        assert parsed.logFormatVariables.get("Type")
        assert parsed.logFormatVariables.get("Content")
        if parsed.EventID > 0:
            n_match += 1
            assert parsed.template == AUDIT_TEMPLATES[parsed.EventID - 1]
            assert parsed.variables  # wildcards captured
    assert n_match > 150  # most generated lines match their template


def test_matcher_parser_batch_equals_single():
    parser = MatcherParser({"templates": AUDIT_TEMPLATES})
    gen = AuditLogGenerator(seed=11)
    frames = [LogSchema(logID=str(i), log=gen.line()[0]).serialize() for i in range(20)]
    singles = [parser.process(f) for f in frames]
    batched = parser.process_batch(frames)
    for s, b in zip(singles, batched):
        ps, pb = ParserSchema.deserialize(s), ParserSchema.deserialize(b)
        assert ps.EventID == pb.EventID
        assert ps.variables == pb.variables


def test_normalization_params():
    parser = MatcherParser(
        {
            "templates": ["hello <*> world"],
            "params": {"lowercase": True},
        }
    )
    frame = LogSchema(log="HELLO Big WORLD").serialize()
    parsed = ParserSchema.deserialize(parser.process(frame))
    assert parsed.EventID == 1
    assert parsed.variables == ["big"]


def test_templates_from_file(tmp_path):
    tf = tmp_path / "templates.txt"
    tf.write_text("alpha=<*>\nbeta=<*> gamma=<*>\n")
    parser = MatcherParser({"params": {"path_templates": str(tf)}})
    parsed = ParserSchema.deserialize(
        parser.process(LogSchema(log="beta=1 gamma=2").serialize())
    )
    assert parsed.EventID == 2
    assert parsed.variables == ["1", "2"]


def test_dummy_parser_contract():
    p = DummyParser()
    parsed = ParserSchema.deserialize(
        p.process(LogSchema(logID="id9", log="whatever").serialize())
    )
    assert parsed.EventID == 2
    assert parsed.template == "This is a dummy template"
    assert parsed.variables == ["dummy_variable"]
    assert parsed.log == "DummyParser"
    assert parsed.logID == "id9"


def test_time_format_extraction():
    """time_format parses the header time into receivedTimestamp
    (reference config key, parser_config.yaml:5)."""
    parser = MatcherParser({
        "log_format": '<IP> - - [<Time>] "<Req>"',
        "time_format": "%d/%b/%Y:%H:%M:%S",
        "templates": [],
    })
    frame = LogSchema(
        log='1.2.3.4 - - [10/Oct/2000:13:55:36] "GET /x"'
    ).serialize()
    parsed = ParserSchema.deserialize(parser.process(frame))
    assert parsed.logFormatVariables["Time"] == "10/Oct/2000:13:55:36"
    import datetime
    expect = int(datetime.datetime(2000, 10, 10, 13, 55, 36,
                                   tzinfo=datetime.timezone.utc).timestamp())
    assert parsed.receivedTimestamp == expect


def test_drift_refit_appends_templates():
    """auto_config_refit_lines: unmatched lines accumulate and are mined
    into NEW templates appended after the existing ones (stable EventIDs)."""
    from detectmateservice_amd.library.parsers import MatcherParser
    from detectmateservice_amd.schemas import LogSchema, ParserSchema

    parser = MatcherParser({
        "templates": ["session opened for user <*> by uid <*>"],
        "auto_config_refit_lines": 24,
    })
    # known pattern parses as event 1
    known = LogSchema(logID="k", log="session opened for user root by uid 0")
    out = ParserSchema.deserialize(parser.process(known.serialize()))
    assert out.EventID == 1

    # a NEW recurring pattern the template set doesn't know
    drift = [
        LogSchema(logID=f"d{i}", log=f"connection from 10.0.0.{i} port {4000+i} accepted").serialize()
        for i in range(30)
    ]
    outs = [ParserSchema.deserialize(o) for o in parser._process_python(drift)]
    assert all(o.EventID == -1 for o in outs[:20])  # unmatched before refit
    assert len(parser.templates) > 1  # refit appended a mined template

    # the drifted pattern now matches with a NEW id; event 1 is unchanged
    again = LogSchema(logID="d", log="connection from 10.0.0.99 port 4099 accepted")
    o2 = ParserSchema.deserialize(parser.process(again.serialize()))
    assert o2.EventID > 1
    o1 = ParserSchema.deserialize(parser.process(known.serialize()))
    assert o1.EventID == 1


def test_drift_refit_batched_path():
    """Same behavior through the batched C++/kernel path."""
    from detectmateservice_amd import ops
    from detectmateservice_amd.library.parsers import MatcherParser
    from detectmateservice_amd.schemas import LogSchema, ParserSchema

    if not ops.have_extension():
        pytest.skip("extension not built")
    parser = MatcherParser({
        "templates": ["session opened for user <*> by uid <*>"],
        "auto_config_refit_lines": 24,
    })
    drift = [
        LogSchema(logID=f"d{i}", log=f"connection from 10.0.0.{i} port {4000+i} accepted").serialize()
        for i in range(30)
    ]
    parser.process_batch(drift)
    assert len(parser.templates) > 1
    again = [LogSchema(logID="x", log="connection from 10.0.0.99 port 4099 accepted").serialize()] * 8
    outs = [ParserSchema.deserialize(o) for o in parser.process_batch(again)]
    assert all(o.EventID > 1 for o in outs)


def test_template_aging_on_refit():
    """A template that matches nothing for template_max_idle_lines is
    retired at the next drift refit: its EventID stops being assigned
    (never reused), live templates keep their positional ids, and newly
    mined drift templates append after it."""
    from detectmateservice_amd.library.parsers import MatcherParser
    from detectmateservice_amd.schemas import LogSchema

    parser = MatcherParser({
        "method_type": "matcher_parser",
        "templates": ["alive <*> ok", "deadwood <*> gone"],
        "auto_config_refit_lines": 8,
        "template_max_idle_lines": 10,
    })

    def frames(lines):
        return [LogSchema(logID=str(i), log=l).serialize()
                for i, l in enumerate(lines)]

    import json

    from detectmateservice_amd.schemas import ParserSchema

    # template 2 matches once up front: its idle clock starts at 0
    out = parser.process_batch(frames(["deadwood 1 gone"]))
    assert ParserSchema.deserialize(out[0]).EventID == 2

    # 12 lines that only exercise template 1 -> template 2 idles past 10
    out = parser.process_batch(frames([f"alive {i} ok" for i in range(12)]))
    assert all(ParserSchema.deserialize(o).EventID == 1 for o in out)

    # 8 unmatched lines trigger the drift refit, which ages template 2
    out = parser.process_batch(frames([f"novel thing {i} seen" for i in range(8)]))
    assert parser._active == [True, False] + [True] * (len(parser.templates) - 2)
    assert len(parser.templates) > 2  # drift mined something new

    # retired id 2 no longer matches; id 1 still does
    out = parser.process_batch(frames(["deadwood 9 gone", "alive 9 ok"]))
    evs = [ParserSchema.deserialize(o).EventID for o in out]
    assert evs[1] == 1
    assert evs[0] != 2  # -1 or a new drift template, never the retired id
