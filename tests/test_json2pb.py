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
