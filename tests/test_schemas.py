"""Wire-schema tests: roundtrip + wire compatibility against the official
protobuf runtime (the reference's fluentd plugins speak this exact format —
SURVEY.md §2.3)."""
import pytest

from detectmateservice_amd.schemas import (
    DetectorSchema,
    LogSchema,
    OutputSchema,
    ParserSchema,
)


def test_log_schema_roundtrip():
    s = LogSchema(logID="abc-1", log="hello world", logSource="file", hostname="h1")
    data = s.serialize()
    back = LogSchema.deserialize(data)
    assert back == s
    assert back.log == "hello world"


def test_parser_schema_roundtrip_full():
    s = ParserSchema(
        parserType="matcher_parser",
        parserID="p1",
        EventID=7,
        template="pid=<*> uid=<*>",
        variables=["123", "0"],
        parsedLogID="pl1",
        logID="l1",
        log="pid=123 uid=0",
        logFormatVariables={"Type": "LOGIN", "Time": "123.456"},
        receivedTimestamp=1700000000,
        parsedTimestamp=1700000001,
    )
    back = ParserSchema.deserialize(s.serialize())
    assert back == s
    assert back.variables == ["123", "0"]
    assert back.logFormatVariables["Type"] == "LOGIN"


def test_detector_schema_roundtrip():
    s = DetectorSchema(
        detectorID="d1",
        detectorType="new_value_detector",
        alertID="a1",
        detectionTimestamp=1700000002,
        logIDs=["l1", "l2"],
        score=0.75,
        extractedTimestamps=[1, 2, 3],
        description="Unknown value: '/foobar'",
        receivedTimestamp=5,
        alertsObtain={"Global - URL": "/foobar"},
    )
    back = DetectorSchema.deserialize(s.serialize())
    assert back == s
    assert back.score == pytest.approx(0.75)
    assert back.extractedTimestamps == [1, 2, 3]


def test_output_schema_roundtrip():
    s = OutputSchema(
        detectorIDs=["d1"],
        detectorTypes=["nvd"],
        alertIDs=["a1"],
        outputTimestamp=9,
        logIDs=["l1"],
        description="agg",
    )
    assert OutputSchema.deserialize(s.serialize()) == s


def test_defaults_not_emitted():
    assert LogSchema(__version__="").serialize() != b""  # version auto-set
    s = LogSchema()
    # only the version field should be on the wire
    data = s.serialize()
    assert len(data) < 16


def test_negative_int32_roundtrip():
    s = ParserSchema(EventID=-1)
    back = ParserSchema.deserialize(s.serialize())
    assert back.EventID == -1


def test_unknown_fields_skipped():
    from detectmateservice_amd.schemas import codec

    # encode with an extra field number 99 (varint) prepended
    payload = codec.encode_tag(99, 0) + codec.encode_varint(42)
    payload += LogSchema(log="x").serialize()
    back = LogSchema.deserialize(payload)
    assert back.log == "x"


def _build_pb2_parser_schema():
    """Build the same ParserSchema via google.protobuf dynamic messages."""
    from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

    pool = descriptor_pool.DescriptorPool()
    f = descriptor_pb2.FileDescriptorProto()
    f.name = "schemas_test.proto"
    f.package = "dmtest"
    f.syntax = "proto3"
    m = f.message_type.add()
    m.name = "ParserSchema"

    def add(name, number, ftype, label=1, type_name=None):
        fld = m.field.add()
        fld.name = name
        fld.number = number
        fld.type = ftype
        fld.label = label
        if type_name:
            fld.type_name = type_name

    T = descriptor_pb2.FieldDescriptorProto
    add("version", 1, T.TYPE_STRING)
    add("parserType", 2, T.TYPE_STRING)
    add("parserID", 3, T.TYPE_STRING)
    add("EventID", 4, T.TYPE_INT32)
    add("template", 5, T.TYPE_STRING)
    add("variables", 6, T.TYPE_STRING, label=3)
    add("parsedLogID", 7, T.TYPE_STRING)
    add("logID", 8, T.TYPE_STRING)
    add("log", 9, T.TYPE_STRING)
    # map<string,string> = repeated nested MapEntry
    entry = m.nested_type.add()
    entry.name = "LogFormatVariablesEntry"
    entry.options.map_entry = True
    k = entry.field.add(); k.name = "key"; k.number = 1; k.type = T.TYPE_STRING; k.label = 1
    v = entry.field.add(); v.name = "value"; v.number = 2; v.type = T.TYPE_STRING; v.label = 1
    add("logFormatVariables", 10, T.TYPE_MESSAGE, label=3,
        type_name=".dmtest.ParserSchema.LogFormatVariablesEntry")
    add("receivedTimestamp", 11, T.TYPE_INT32)
    add("parsedTimestamp", 12, T.TYPE_INT32)

    pool.Add(f)
    desc = pool.FindMessageTypeByName("dmtest.ParserSchema")
    return message_factory.GetMessageClass(desc)


def test_wire_compat_with_official_protobuf():
    """Our codec's bytes must decode identically via google.protobuf and
    vice versa (fluentd interop guarantee)."""
    Pb = _build_pb2_parser_schema()

    ours = ParserSchema(
        parserType="matcher_parser",
        EventID=5,
        variables=["a", "b"],
        log="line",
        logFormatVariables={"Type": "LOGIN"},
        receivedTimestamp=1700000000,
    )
    msg = Pb()
    msg.ParseFromString(ours.serialize())
    assert msg.parserType == "matcher_parser"
    assert msg.EventID == 5
    assert list(msg.variables) == ["a", "b"]
    assert msg.log == "line"
    assert dict(msg.logFormatVariables) == {"Type": "LOGIN"}
    assert msg.receivedTimestamp == 1700000000

    # reverse direction
    msg2 = Pb()
    msg2.version = "0.3"
    msg2.parserType = "x"
    msg2.EventID = -3
    msg2.variables.extend(["q"])
    msg2.logFormatVariables["k"] = "v"
    back = ParserSchema.deserialize(msg2.SerializeToString())
    assert back.parserType == "x"
    assert back.EventID == -3
    assert back.variables == ["q"]
    assert back.logFormatVariables == {"k": "v"}
