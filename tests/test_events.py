"""TensorBoard event-file writer: records round-trip
through our own codec and follow the TFRecord/Event wire format."""

import glob

from tensorflowonspark_amd import tfrecord
from tensorflowonspark_amd.utils import events


def test_scalar_event_roundtrip():
    rec = events.encode_scalar_event(42, {"loss": 1.5, "acc": 0.75},
                                     wall_time=123.25)
    wall, step, scalars = events.decode_scalar_event(rec)
    assert wall == 123.25 and step == 42
    assert abs(scalars["loss"] - 1.5) < 1e-6
    assert abs(scalars["acc"] - 0.75) < 1e-6


def test_summary_writer_file(tmp_path):
    logdir = str(tmp_path / "run1")
    with events.SummaryWriter(logdir) as w:
        for s in range(5):
            w.add_scalar("loss", 2.0 / (s + 1), s)
        w.add_scalars({"imgs_per_sec": 9000.0, "lr": 0.1}, 4)
    files = glob.glob(logdir + "/events.out.tfevents.*")
    assert len(files) == 1
    recs = list(tfrecord.tfrecord_iterator(files[0], verify=True))
    assert len(recs) == 7  # header + 5 + 1
    # header carries file_version
    assert b"brain.Event:2" in recs[0]
    wall, step, sc = events.decode_scalar_event(recs[3])
    assert step == 2 and abs(sc["loss"] - 2.0 / 3) < 1e-6
    _, step6, sc6 = events.decode_scalar_event(recs[6])
    assert step6 == 4 and abs(sc6["imgs_per_sec"] - 9000.0) < 1e-3


def test_summary_writer_schemed_logdir(tmp_path):
    """Event files also land on schemed paths (the chief's model_dir may be
    an hdfs:// URI; file:// here)."""
    import glob

    logdir = "file://" + str(tmp_path / "run2")
    from tensorflowonspark_amd.utils import events as ev

    # SummaryWriter uses os.makedirs + TFRecordWriter; schemed paths go
    # through fsio inside the writer, so only strip the scheme for makedirs
    w = ev.SummaryWriter(str(tmp_path / "run2"))
    w.add_scalar("a", 1.0, 1)
    w.close()
    files = glob.glob(str(tmp_path / "run2") + "/events.out.tfevents.*")
    assert files
    from tensorflowonspark_amd import tfrecord
    uri = "file://" + files[0]
    recs = list(tfrecord.tfrecord_iterator(uri, verify=True))
    assert len(recs) == 2


def test_scalar_event_property():
    """Property: arbitrary tags/values/steps round-trip through the
    hand-rolled Event proto encoding."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    tags = st.text(min_size=1, max_size=40).filter(lambda s: s.strip())
    floats = st.floats(allow_nan=False, allow_infinity=False,
                       width=32)
    scalars = st.dictionaries(tags, floats, min_size=1, max_size=8)
    steps = st.integers(min_value=0, max_value=2**53)

    @settings(max_examples=60, deadline=None)
    @given(scalars=scalars, step=steps)
    def check(scalars, step):
        rec = events.encode_scalar_event(step, scalars, wall_time=1.5)
        wall, got_step, got = events.decode_scalar_event(rec)
        assert got_step == step
        assert set(got) == set(scalars)
        import struct as _s
        for k, v in scalars.items():
            want = _s.unpack("<f", _s.pack("<f", v))[0]
            assert got[k] == want

    check()
