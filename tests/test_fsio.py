"""Scheme-aware I/O: file:// and memory:// (fsspec) TFRecord round trips.

VERDICT r01 item 4: `hdfs_path` outputs must be openable — dfutil round-trip
parameterized over a file://-schemed URI and an fsspec in-memory filesystem
(the reference reached HDFS via tensorflow-hadoop, dfutil.py:39-41).
"""

import pytest

from tensorflowonspark_amd import dfutil, tfrecord
from tensorflowonspark_amd.local_context import LocalSparkContext
from tensorflowonspark_amd.utils import dataset, fsio


def test_get_scheme():
    assert fsio.get_scheme("hdfs://nn:8020/user/x") == "hdfs"
    assert fsio.get_scheme("file:///tmp/x") == "file"
    assert fsio.get_scheme("memory://bucket/x") == "memory"
    assert fsio.get_scheme("/tmp/plain") is None
    assert fsio.get_scheme("relative/path") is None


def test_tfrecord_roundtrip_file_scheme(tmp_path):
    uri = "file://" + str(tmp_path / "recs" / "part-r-00000")
    with tfrecord.TFRecordWriter(uri) as w:
        for i in range(5):
            w.write(tfrecord.encode_example({"x": [i], "s": [b"v%d" % i]}))
    recs = list(tfrecord.tfrecord_iterator(uri, verify=True))
    assert len(recs) == 5
    ex = tfrecord.decode_example(recs[3])
    assert ex["x"][1] == [3]


def test_tfrecord_roundtrip_memory_scheme():
    pytest.importorskip("fsspec")
    uri = "memory://tfosr_test/part-r-00000"
    with tfrecord.TFRecordWriter(uri) as w:
        for i in range(4):
            w.write(tfrecord.encode_example({"v": [float(i)]}))
    recs = list(tfrecord.tfrecord_iterator(uri, verify=True))
    assert len(recs) == 4


def test_shard_files_schemed_dir(tmp_path):
    d = tmp_path / "shards"
    d.mkdir()
    for i in range(6):
        (d / "part-r-{:05d}".format(i)).write_bytes(b"")
    uri = "file://" + str(d)
    files = dataset.shard_files(uri, 0, 2)
    assert len(files) == 3
    assert all(f.startswith("file://") for f in files)
    # both shards together cover all 6
    files2 = dataset.shard_files(uri, 1, 2)
    assert len(set(files) | set(files2)) == 6


def test_dfutil_roundtrip_file_scheme(tmp_path):
    sc = LocalSparkContext(2)
    df = sc.createDataFrame(
        [(1, 2.5, "a"), (2, 3.5, "b"), (3, 4.5, "c")],
        ["i", "f", "s"], ["bigint", "double", "string"])
    out = "file://" + str(tmp_path / "tfr_out")
    dfutil.saveAsTFRecords(df, out)
    df2 = dfutil.loadTFRecords(sc, out)
    assert dfutil.isLoadedDF(df2)
    rows = sorted(df2.collect())
    assert rows[0][:2] == (2.5, 1) or rows[0][0] in (1, 2.5)
    # column order is name-sorted on load: f, i, s
    assert [r[1] for r in rows] == [1, 2, 3]
    assert [r[0] for r in rows] == [2.5, 3.5, 4.5]
    assert [r[2] for r in rows] == ["a", "b", "c"]


def test_fs_glob_and_listfiles_memory_scheme():
    # memory:// is per-process, so exercise the listing layer in-process
    # (LocalSparkContext partitions run in child processes; the distributed
    # round trip over a shared filesystem is test_dfutil_roundtrip_file_scheme)
    pytest.importorskip("fsspec")
    base = "memory://tfosr_listing"
    for i in range(3):
        with tfrecord.TFRecordWriter(base + "/part-r-{:05d}".format(i)) as w:
            w.write(tfrecord.encode_example({"v": [i]}))
    globbed = fsio.fs_glob(base + "/part-*")
    assert len(globbed) == 3
    assert all(g.startswith("memory://") for g in globbed)
    listed = fsio.fs_listfiles(base)
    assert listed == globbed
    # shard_files accepts the schemed dir and the results re-open
    files = dataset.shard_files(base, 0, 1)
    assert len(files) == 3
    assert len(list(tfrecord.tfrecord_iterator(files[0], verify=True))) == 1
