"""mcpack v2 codec (reference mcpack2pb/field_type.h + serializer.cpp head
layouts, clean-room): schema-less value tree with the exact binary field
heads (FieldFixedHead/ShortHead/LongHead + ItemsHead), plus a JSON bridge
standing in for the protoc-gen-mcpack front-end."""
import struct

import brpc_amd as b
import pytest

c = b.core.codecs


def test_roundtrip_scalars():
    doc = {"i8": 5, "neg": -3, "i32": 100000, "i64": 1 << 40, "f": 3.5,
           "t": True, "f2": False, "none": None, "s": "hello", "raw": b"\x00\x01\xff"}
    blob = c.mcpack_dumps(doc)
    back = c.mcpack_loads(blob)
    assert back == doc


def test_roundtrip_nested():
    doc = {"obj": {"a": 1, "b": {"c": [1, 2, 3], "d": "x"}},
           "arr": [{"k": "v"}, [True, None], "end"]}
    blob = c.mcpack_dumps(doc)
    assert c.mcpack_loads(blob) == doc


def test_long_string_and_binary():
    doc = {"s": "x" * 1000, "bin": b"y" * 1000}
    assert c.mcpack_loads(c.mcpack_dumps(doc)) == doc


def test_exact_wire_layout_small_int():
    # {"a": 5} -> object field: FieldLongHead(0x10, name 0, vsize u32)
    #   ItemsHead(1) + FieldFixedHead(0x11 int8, name "a\0") + value 5
    blob = c.mcpack_dumps({"a": 5})
    assert blob[0] == 0x10  # FIELD_OBJECT
    assert blob[1] == 0  # unnamed root
    (vsize,) = struct.unpack_from("<I", blob, 2)
    assert vsize == len(blob) - 6
    (count,) = struct.unpack_from("<I", blob, 6)
    assert count == 1
    assert blob[10] == 0x11  # FIELD_INT8
    assert blob[11] == 2  # name_size counts the '\0'
    assert blob[12:14] == b"a\x00"
    assert blob[14] == 5


def test_exact_wire_layout_short_string():
    blob = c.mcpack_dumps({"s": "hi"})
    # field head at offset 10: short string head 0x50|0x80, name 2, vsize 3
    assert blob[10] == 0xD0
    assert blob[11] == 2
    assert blob[12] == 3  # "hi\0"
    assert blob[13:15] == b"s\x00"
    assert blob[15:18] == b"hi\x00"


def test_parse_isoarray():
    # hand-crafted: {"v": ISOARRAY int32 [7, 9]}
    items = struct.pack("<B", 0x14) + struct.pack("<ii", 7, 9)
    field = bytes([0x30, 2]) + struct.pack("<I", len(items)) + b"v\x00" + items
    body = struct.pack("<I", 1) + field
    blob = bytes([0x10, 0]) + struct.pack("<I", len(body)) + body
    assert c.mcpack_loads(blob) == {"v": [7, 9]}


def test_to_json():
    j = c.mcpack_to_json(c.mcpack_dumps({"a": 1, "s": "x", "n": None, "l": [True]}))
    assert j == '{"a":1,"l":[true],"n":null,"s":"x"}'


def test_corrupt_rejected():
    with pytest.raises(RuntimeError):
        c.mcpack_loads(b"\x10\x00\xff\xff\xff\xff")


def test_isoarray_roundtrip_and_layout():
    """Uniform int arrays serialize as ISOARRAY (0x30 + IsoItemsHead) and
    parse back identically."""
    blob = c.mcpack_dumps({"v": [7, -9, 100000]})
    assert c.mcpack_loads(blob) == {"v": [7, -9, 100000]}
    # field head at offset 10: ISOARRAY, name "v\0", vsize = 1 + 3*4
    assert blob[10] == 0x30
    (vsize,) = struct.unpack_from("<I", blob, 12)
    assert vsize == 13
    assert blob[16 + 2] == 0x14  # item type INT32 after name
