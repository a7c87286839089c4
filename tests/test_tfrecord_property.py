"""Property-based tests for the TFRecord codec (hypothesis)."""

import hypothesis.strategies as st
import pytest
from hypothesis import HealthCheck, given, settings

from tensorflowonspark_amd import tfrecord

names = st.text(alphabet=st.characters(min_codepoint=33, max_codepoint=126),
                min_size=1, max_size=12)
int_lists = st.lists(st.integers(min_value=-2**62, max_value=2**62 - 1),
                     min_size=1, max_size=8)
float_lists = st.lists(st.floats(width=32, allow_nan=False,
                                 allow_infinity=False),
                       min_size=1, max_size=8)
bytes_lists = st.lists(st.binary(max_size=32), min_size=1, max_size=4)


@settings(max_examples=120, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(st.dictionaries(names, st.one_of(int_lists, float_lists, bytes_lists),
                       min_size=1, max_size=6))
def test_example_roundtrip(feats):
    rec = tfrecord.encode_example(feats)
    decoded = tfrecord.decode_example(rec)
    assert sorted(decoded.keys()) == sorted(feats.keys())
    for name, vals in feats.items():
        kind, got = decoded[name]
        if isinstance(vals[0], bytes):
            assert kind == "bytes" and got == vals
        elif isinstance(vals[0], int):
            assert kind == "int64" and got == vals
        else:
            assert kind == "float"
            assert got == pytest.approx(vals, rel=1e-6, abs=1e-30)


@settings(max_examples=60, deadline=None)
@given(st.lists(st.binary(max_size=256), min_size=0, max_size=12))
def test_framing_roundtrip(records):
    import os
    import tempfile
    path = os.path.join(tempfile.mkdtemp(), "f.tfrecord")
    with tfrecord.TFRecordWriter(path) as w:
        for rec in records:
            w.write(rec)
    assert list(tfrecord.tfrecord_iterator(path, verify=True)) == records


@settings(max_examples=60, deadline=None)
@given(st.binary(min_size=0, max_size=512))
def test_crc32c_matches_native(data):
    """Python table CRC vs the C++ SSE4.2 implementation (when built)."""
    from tensorflowonspark_amd.ops import get_ext
    ext = get_ext(required=False)
    if ext is None or not hasattr(ext, "crc32c"):
        pytest.skip("native codec not built")
    assert tfrecord.crc32c(data) == ext.crc32c(data)
