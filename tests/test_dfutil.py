"""TFRecord codec + dfutil round-trip tests
(shape parity: reference tests/test_dfutil.py)."""

import pytest

from tensorflowonspark_amd import dfutil, tfrecord
from tensorflowonspark_amd.local_context import LocalSparkContext


def test_crc32c_vector():
    assert tfrecord.crc32c(b"123456789") == 0xE3069283
    assert tfrecord.crc32c(b"") == 0


def test_example_all_types_roundtrip():
    feats = {
        "f_int": 42, "f_neg": -7, "f_bool": True, "f_float": 2.5,
        "f_str": "hello", "f_bytes": b"\x01\x02",
        "a_int": [1, 2, 3], "a_float": [0.5, 1.5], "a_str": ["a", "b"],
    }
    d = tfrecord.decode_example(tfrecord.encode_example(feats))
    assert d["f_int"] == ("int64", [42])
    assert d["f_neg"] == ("int64", [-7])
    assert d["f_bool"] == ("int64", [1])
    assert d["f_float"][1][0] == pytest.approx(2.5)
    assert d["f_str"] == ("bytes", [b"hello"])
    assert d["f_bytes"] == ("bytes", [b"\x01\x02"])
    assert d["a_int"] == ("int64", [1, 2, 3])
    assert [pytest.approx(v) for v in d["a_float"][1]] == [0.5, 1.5]
    assert d["a_str"] == ("bytes", [b"a", b"b"])


@pytest.fixture()
def sc():
    ctx = LocalSparkContext(num_executors=2)
    yield ctx
    ctx.stop()


def test_df_tfrecord_roundtrip(sc, tmp_path):
    rows = [(i, float(i) * 0.5, "s{}".format(i), [1.0 * i, 2.0 * i])
            for i in range(20)]
    df = sc.createDataFrame(rows, ["idx", "val", "name", "vec"])
    assert dict(df.dtypes)["vec"] == "array<double>"

    out = str(tmp_path / "tfr")
    dfutil.saveAsTFRecords(df, out)

    df2 = dfutil.loadTFRecords(sc, out)
    assert dfutil.isLoadedDF(df2)
    assert df2.count() == 20
    # columns come back sorted by name
    assert df2.columns == ["idx", "name", "val", "vec"]
    got = {r[0]: r for r in df2.collect()}
    assert got[3][1] == "s3"
    assert got[3][2] == pytest.approx(1.5)
    assert got[3][3] == [pytest.approx(3.0), pytest.approx(6.0)]


def test_binary_features_hint(sc, tmp_path):
    rows = [(b"\x00\xff", "text")]
    df = sc.createDataFrame(rows, ["blob", "txt"])
    out = str(tmp_path / "tfr2")
    dfutil.saveAsTFRecords(df, out)
    df2 = dfutil.loadTFRecords(sc, out, binary_features=["blob"])
    r = df2.collect()[0]
    assert r[0] == b"\x00\xff"   # stays bytes
    assert r[1] == "text"        # decoded to str


def test_load_with_schema_hint(tmp_path):
    """schema_hint overrides inference and preserves hinted field order
    (reference DFUtil.scala:35-55,67-110)."""
    from tensorflowonspark_amd import dfutil
    from tensorflowonspark_amd.local_context import LocalSparkContext
    sc = LocalSparkContext(2)
    df = sc.createDataFrame([(1, 2.5, "x"), (2, 3.5, "y")],
                            ["num", "val", "txt"],
                            ["bigint", "double", "string"])
    out = str(tmp_path / "recs")
    dfutil.saveAsTFRecords(df, out)
    # hint: txt first, and store it as binary instead of string
    df2 = dfutil.loadTFRecords(sc, out,
                               schema_hint="struct<txt:binary,num:bigint>")
    assert df2.columns[0] == "txt" and df2.columns[1] == "num"
    rows = sorted(df2.collect(), key=lambda r: r[1])
    assert rows[0][1] == 1 and isinstance(rows[0][0], (bytes, bytearray))
    assert rows[0][2] == 2.5  # unhinted column keeps inferred dtype
