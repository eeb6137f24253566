"""Property-based tests (hypothesis): model-check the byte-hot paths
against trivially-correct python references, the way the reference's
iobuf_unittest exhausts op interleavings (test/iobuf_unittest.cpp) but
with generated programs.

- IOBuf: random op sequences vs a plain python bytes model
- snappy: compress/decompress round-trip on adversarial byte shapes
- HPACK: header-list round-trip (Huffman + tables) on arbitrary strings
- crc32c: Combine vs whole-buffer over random split points
- mcpack <-> json: round-trip on generated object trees
"""
import json

import brpc_amd as b
from hypothesis import given, settings, strategies as st

MAX_EXAMPLES = 60


@settings(max_examples=MAX_EXAMPLES, deadline=None)
@given(st.lists(
    st.one_of(
        st.tuples(st.just("append"), st.binary(min_size=0, max_size=5000)),
        st.tuples(st.just("cutn"), st.integers(0, 6000)),
        st.tuples(st.just("pop_front"), st.integers(0, 6000)),
        st.tuples(st.just("pop_back"), st.integers(0, 6000)),
    ),
    max_size=40))
def test_iobuf_model(ops):
    buf = b.core.IOBuf()
    model = b""
    for op, arg in ops:
        if op == "append":
            buf.append(arg)
            model += arg
        elif op == "cutn":
            got = buf.cutn(arg)
            want, model = model[:arg], model[arg:]
            assert got == want
        elif op == "pop_front":
            n = min(arg, len(model))
            buf.pop_front(arg)
            model = model[n:]
        elif op == "pop_back":
            n = min(arg, len(model))
            buf.pop_back(arg)
            model = model[:len(model) - n]
        assert buf.size() == len(model)
    assert buf.to_bytes() == model


@settings(max_examples=MAX_EXAMPLES, deadline=None)
@given(st.binary(min_size=0, max_size=200000))
def test_snappy_roundtrip(data):
    comp = b.core.snappy.compress(data)
    assert b.core.snappy.uncompress(comp) == data


@settings(max_examples=MAX_EXAMPLES, deadline=None)
@given(st.lists(st.tuples(
    st.text(alphabet=st.characters(min_codepoint=0x21, max_codepoint=0x7e),
            min_size=1, max_size=40).map(str.lower),
    st.text(alphabet=st.characters(min_codepoint=0x20, max_codepoint=0xff),
            min_size=0, max_size=200)),
    min_size=0, max_size=20))
def test_hpack_roundtrip(headers):
    enc = b.core.hpack.Encoder()
    dec = b.core.hpack.Decoder()
    # two blocks through the SAME tables: dynamic-table state must agree
    expected = [(k.encode(), v.encode()) for k, v in headers]
    for _ in range(2):
        wire = enc.encode(headers)
        out = dec.decode(wire)
        assert [tuple(h) for h in out] == expected


@settings(max_examples=MAX_EXAMPLES, deadline=None)
@given(st.binary(min_size=1, max_size=50000), st.data())
def test_crc32c_combine(data, dd):
    cut = dd.draw(st.integers(0, len(data)))
    whole = b.core.crc32c(data)
    a, bb = data[:cut], data[cut:]
    combined = b.core.crc32c_combine(b.core.crc32c(a), b.core.crc32c(bb), len(bb))
    assert combined == whole


json_values = st.recursive(
    st.one_of(st.none(), st.booleans(),
              st.integers(min_value=-2**31, max_value=2**31 - 1),
              st.text(max_size=30)),
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(min_size=1, max_size=10), children, max_size=4)),
    max_leaves=12)


@settings(max_examples=MAX_EXAMPLES, deadline=None)
@given(st.dictionaries(st.text(min_size=1, max_size=10), json_values, max_size=5))
def test_mcpack_json_roundtrip(obj):
    blob = b.core.codecs.mcpack_dumps(obj)
    back = b.core.codecs.mcpack_loads(blob)
    assert back == obj
