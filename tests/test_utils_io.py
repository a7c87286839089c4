"""Dataset sharding (InputMode.TENSORFLOW path), checkpoint layout, metrics,
plus a full-cluster TENSORFLOW-mode training run."""

import os

import pytest
import torch

from tensorflowonspark_amd import TFCluster, tfrecord
from tensorflowonspark_amd.local_context import LocalSparkContext
from tensorflowonspark_amd.utils import checkpoint as ckpt
from tensorflowonspark_amd.utils.dataset import TFRecordDataset, shard_files
from tensorflowonspark_amd.utils.metrics import StepTimer


def _write_tfrecords(dirpath, nfiles=4, per_file=25):
    os.makedirs(dirpath, exist_ok=True)
    n = 0
    for f in range(nfiles):
        with tfrecord.TFRecordWriter(
                os.path.join(dirpath, "part-r-{:05d}".format(f))) as w:
            for _ in range(per_file):
                w.write(tfrecord.encode_example(
                    {"x": [float(n) / 100.0], "y": [2.0 * n / 100.0]}))
                n += 1
    return n


def test_shard_files(tmp_path):
    _write_tfrecords(str(tmp_path), nfiles=5)
    s0 = shard_files(str(tmp_path), 0, 2)
    s1 = shard_files(str(tmp_path), 1, 2)
    assert len(s0) == 3 and len(s1) == 2
    assert not set(s0) & set(s1)


def test_tfrecord_dataset(tmp_path):
    total = _write_tfrecords(str(tmp_path))
    ds = TFRecordDataset(str(tmp_path), 0, 1, batch_size=32)
    assert ds.count() == total
    rows = 0
    for batch in ds:
        assert len(batch) <= 32
        assert "x" in batch[0]
        rows += len(batch)
    assert rows == total
    # two epochs re-iterate cleanly
    assert sum(len(b) for b in ds) == total


def test_checkpoint_roundtrip(tmp_path):
    model = torch.nn.Linear(3, 2)
    md = str(tmp_path / "model_dir")
    assert ckpt.latest_checkpoint(md) is None
    for step in (1, 2, 7):
        ckpt.save_checkpoint(md, step, model, optimizer_state={"lr": 0.1 * step})
    assert ckpt.latest_checkpoint(md).endswith("weights-0007.pt")

    model2 = torch.nn.Linear(3, 2)
    step, opt_state = ckpt.load_latest(md, model2)
    assert step == 7 and opt_state["lr"] == pytest.approx(0.7)
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p1, p2)


def test_checkpoint_keep_last(tmp_path):
    model = torch.nn.Linear(2, 2)
    md = str(tmp_path / "md")
    for step in range(8):
        ckpt.save_checkpoint(md, step, model, keep_last=3)
    import glob
    assert len(glob.glob(os.path.join(md, "weights-*.pt"))) == 3


def test_step_timer():
    timer = StepTimer(batch_size=10, log_every=2)
    timer.start()
    assert timer.step() is None
    ips = timer.step()
    assert ips is not None and ips > 0


def _direct_read_fn(args, ctx):
    """TENSORFLOW-mode map_fun: read the shard directly, fit y=2x."""
    import torch

    from tensorflowonspark_amd.utils.dataset import TFRecordDataset
    workers = len(ctx.cluster_spec.get("worker", [])) or 1
    ds = TFRecordDataset(args["data_dir"], ctx.task_index, workers, batch_size=16)
    model = torch.nn.Linear(1, 1, bias=False)
    opt = torch.optim.SGD(model.parameters(), lr=0.5)
    for _epoch in range(30):
        for batch in ds:
            x = torch.tensor([ex["x"][1] for ex in batch])
            y = torch.tensor([ex["y"][1] for ex in batch])
            opt.zero_grad()
            loss = torch.nn.functional.mse_loss(model(x), y)
            loss.backward()
            opt.step()
    with open("weight.txt", "w") as f:
        f.write(str(model.weight.item()))


@pytest.mark.timeout(300)
def test_cluster_tensorflow_mode(tmp_path):
    data_dir = str(tmp_path / "tfr")
    _write_tfrecords(data_dir)
    sc = LocalSparkContext(num_executors=2)
    try:
        cluster = TFCluster.run(sc, _direct_read_fn, {"data_dir": data_dir},
                                num_executors=2, num_ps=0, master_node=None,
                                input_mode=TFCluster.InputMode.TENSORFLOW,
                                num_gpus=0, reservation_timeout=60)
        cluster.shutdown(grace_secs=0)
        import glob
        weights = [float(open(f).read()) for f in
                   glob.glob(os.path.join(sc._root, "executor_*", "weight.txt"))]
        assert len(weights) == 2
        for w in weights:
            assert abs(w - 2.0) < 0.1, weights
    finally:
        sc.stop()


def test_tfrecord_dataset_shuffle(tmp_path):
    total = _write_tfrecords(str(tmp_path), nfiles=2, per_file=30)
    ds = TFRecordDataset(str(tmp_path), 0, 1, batch_size=60, shuffle_buffer=16)
    epoch1 = [ex["x"][1][0] for b in ds for ex in b]
    epoch2 = [ex["x"][1][0] for b in ds for ex in b]
    assert len(epoch1) == len(epoch2) == total
    assert sorted(epoch1) == sorted(epoch2)       # same elements
    assert epoch1 != sorted(epoch1)               # actually shuffled
    assert epoch1 != epoch2                       # reshuffled per epoch


def test_checkpoint_pruning_past_9999(tmp_path):
    """lexicographic pruning broke at step 10000 ('weights-10000'
    sorts before 'weights-9999'); pruning must be numeric."""
    import torch.nn as nn

    from tensorflowonspark_amd.utils import checkpoint as ckpt
    m = nn.Linear(2, 2)
    d = str(tmp_path / "ck")
    for step in (9998, 9999, 10000, 10001, 10002):
        ckpt.save_checkpoint(d, step, m, keep_last=3)
    import glob
    import os
    left = sorted(int(os.path.basename(p).split("-")[1].split(".")[0])
                  for p in glob.glob(d + "/weights-*.pt"))
    assert left == [10000, 10001, 10002], left
    assert ckpt.latest_checkpoint(d).endswith("weights-10002.pt")


def test_hdfs_path_output_openable(tmp_path):
    """hdfs_path's URIs must be openable — with a
    file:// defaultFS the produced URI round-trips through the TFRecord
    writer/reader."""
    from tensorflowonspark_amd import TFNode, tfrecord

    class Ctx:
        defaultFS = "file://"
        working_dir = str(tmp_path)

    uri = TFNode.hdfs_path(Ctx(), "out/part-r-00000")
    assert uri.startswith("file://")
    with tfrecord.TFRecordWriter(uri) as w:
        w.write(tfrecord.encode_example({"v": [7]}))
    recs = list(tfrecord.tfrecord_iterator(uri, verify=True))
    assert len(recs) == 1


def test_tfrecord_dataset_schemed_dir(tmp_path):
    """InputMode.TENSORFLOW reader over a schemed URI directory."""
    from tensorflowonspark_amd import tfrecord
    from tensorflowonspark_amd.utils.dataset import TFRecordDataset
    d = tmp_path / "recs"
    d.mkdir()
    for i in range(2):
        with tfrecord.TFRecordWriter(str(d / "part-{:05d}".format(i))) as w:
            for j in range(5):
                w.write(tfrecord.encode_example({"v": [i * 5 + j]}))
    ds = TFRecordDataset("file://" + str(d), shard_index=0, num_shards=1,
                         batch_size=4)
    seen = []
    for batch in ds:
        for ex in batch:
            seen.append(ex["v"][1][0])
    assert sorted(seen) == list(range(10))
