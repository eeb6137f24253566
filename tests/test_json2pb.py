"""json2pb: schema-driven JSON <-> protobuf wire conversion (parity:
reference src/json2pb). Cross-validated against the python protobuf
runtime via a hand-assembled descriptor-free check: our wire output must
parse with google.protobuf's low-level decoder."""
import json

import pytest

import brpc_amd as b

j = b.core.json2pb

SCHEMA = {
    "name": (1, "string"),
    "id": (2, "int64"),
    "score": (3, "double"),
    "active": (4, "bool"),
    "tags": (5, "repeated string"),
    "nums": (6, "repeated int32"),
    "sub": (7, "message", {"x": (1, "int32"), "y": (2, "string")}),
}


def test_roundtrip():
    doc = {"name": "alice", "id": 123456789012, "score": 2.5, "active": True,
           "tags": ["a", "b"], "nums": [1, -2, 3], "sub": {"x": 7, "y": "inner"}}
    wire = j.json_to_pb(SCHEMA, json.dumps(doc))
    assert isinstance(wire, bytes) and len(wire) > 10
    back = json.loads(j.pb_to_json(SCHEMA, wire))
    assert back == doc


def test_negative_int_varint():
    wire = j.json_to_pb({"v": (1, "int64")}, '{"v": -5}')
    back = json.loads(j.pb_to_json({"v": (1, "int64")}, wire))
    assert back["v"] == -5


def test_unknown_json_fields_ignored():
    wire = j.json_to_pb({"a": (1, "int32")}, '{"a": 1, "zzz": "ignored"}')
    assert json.loads(j.pb_to_json({"a": (1, "int32")}, wire)) == {"a": 1}


def test_interop_with_python_protobuf():
    """Our wire bytes must decode identically with the protobuf runtime."""
    try:
        from google.protobuf.internal import decoder  # noqa
        from google.protobuf import descriptor_pb2
    except ImportError:
        pytest.skip("protobuf python runtime unavailable")
    # Use a well-known message: FileDescriptorProto has field 1 = name
    # (string), field 2 = package (string) — matching wire layout.
    wire = j.json_to_pb({"name": (1, "string"), "package": (2, "string")},
                        '{"name": "f.proto", "package": "pkg"}')
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.MergeFromString(wire)
    assert fdp.name == "f.proto"
    assert fdp.package == "pkg"
    # reverse: protobuf-serialized message decodes through pb_to_json
    fdp2 = descriptor_pb2.FileDescriptorProto(name="x.proto", package="a.b")
    back = json.loads(j.pb_to_json({"name": (1, "string"), "package": (2, "string")},
                                   fdp2.SerializeToString()))
    assert back == {"name": "x.proto", "package": "a.b"}


def test_bad_json_raises():
    with pytest.raises(RuntimeError):
        j.json_to_pb({"a": (1, "int32")}, '{"a": ')


def test_descriptor_driven_json2pb_oracle():
    """Round-2: json2pb works on ANY runtime-parsed .proto via the
    DescriptorPool (≙ reference json2pb operating on pb descriptors),
    cross-checked against python protobuf json_format."""
    src = '''
    syntax = "proto3";
    package d2;
    message Point { double x = 1; double y = 2; }
    message Path { string name = 1; repeated Point points = 2; int64 id = 3; }
    '''
    j = b.core.json2pb
    json_text = ('{"name":"route","id":"987654321","points":'
                 '[{"x":1.5,"y":-2.5},{"x":3.0,"y":4.0}]}')
    wire = j.json_to_pb_proto(src, "d2.Path", json_text)
    # oracle parse
    from google.protobuf import descriptor_pb2, descriptor_pool, message_factory
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "d2.proto"; fdp.package = "d2"; fdp.syntax = "proto3"
    F = descriptor_pb2.FieldDescriptorProto
    pt = fdp.message_type.add(); pt.name = "Point"
    for n, i in [("x", 1), ("y", 2)]:
        f = pt.field.add(); f.name = n; f.number = i; f.type = F.TYPE_DOUBLE
        f.label = F.LABEL_OPTIONAL
    pa = fdp.message_type.add(); pa.name = "Path"
    f = pa.field.add(); f.name = "name"; f.number = 1; f.type = F.TYPE_STRING; f.label = F.LABEL_OPTIONAL
    f = pa.field.add(); f.name = "points"; f.number = 2; f.type = F.TYPE_MESSAGE
    f.label = F.LABEL_REPEATED; f.type_name = ".d2.Point"
    f = pa.field.add(); f.name = "id"; f.number = 3; f.type = F.TYPE_INT64; f.label = F.LABEL_OPTIONAL
    dp = descriptor_pool.DescriptorPool(); dp.Add(fdp)
    M = message_factory.GetMessageClassesForFiles(["d2.proto"], dp)["d2.Path"]
    m = M(); m.ParseFromString(wire)
    assert m.name == "route" and m.id == 987654321
    assert [(p.x, p.y) for p in m.points] == [(1.5, -2.5), (3.0, 4.0)]
    # and back
    out = j.pb_to_json_proto(src, "d2.Path", m.SerializeToString())
    import json as pyjson
    d = pyjson.loads(out)
    assert d["name"] == "route"
    assert d["id"] == "987654321"  # proto3 JSON: int64 as string
    assert len(d["points"]) == 2
