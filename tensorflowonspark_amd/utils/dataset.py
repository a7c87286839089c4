"""Worker-side direct data ingestion — the InputMode.TENSORFLOW equivalent.

In the reference, TENSORFLOW mode meant "TF reads TFRecords from HDFS itself"
(reference ``TFCluster.py:43-46``; example ``mnist_tf_ds.py:41``). Here the
worker reads TFRecord part files (or CSV) directly through the native codec,
sharded across the cluster by executor rank — no feeding job, no queues.
"""

import glob
import os

from .. import tfrecord


def shard_files(paths_or_glob, shard_index, num_shards):
    """Deterministically shard a file list across workers.

    Accepts a directory, a glob, a schemed URI (hdfs://, file://, ...) or an
    explicit list; schemed paths are listed through fsio/fsspec."""
    from . import fsio
    if isinstance(paths_or_glob, str):
        scheme = fsio.get_scheme(paths_or_glob)
        if scheme is not None:
            if fsio.fs_isdir(paths_or_glob):
                files = fsio.fs_glob(paths_or_glob.rstrip("/") + "/part-*") or \
                    fsio.fs_listfiles(paths_or_glob)
            else:
                files = fsio.fs_glob(paths_or_glob)
        elif os.path.isdir(paths_or_glob):
            files = sorted(glob.glob(os.path.join(paths_or_glob, "part-*"))) or \
                sorted(f for f in glob.glob(os.path.join(paths_or_glob, "*"))
                       if os.path.isfile(f))
        else:
            files = sorted(glob.glob(paths_or_glob))
    else:
        files = sorted(paths_or_glob)
    return files[shard_index::num_shards]


def tfrecord_examples(paths, decode=True):
    """Iterate decoded Examples ({name: (kind, values)}) or raw record bytes."""
    for path in paths:
        for rec in tfrecord.tfrecord_iterator(path):
            yield tfrecord.decode_example(rec) if decode else rec


def batched(iterable, batch_size, drop_remainder=False):
    batch = []
    for item in iterable:
        batch.append(item)
        if len(batch) == batch_size:
            yield batch
            batch = []
    if batch and not drop_remainder:
        yield batch


class TFRecordDataset:
    """Epoch-iterable TFRecord reader for one worker's shard.

    Usage inside ``map_fun``::

        ds = TFRecordDataset(ctx.absolute_path(args.data_dir),
                             shard_index=ctx.task_index,
                             num_shards=len(ctx.cluster_spec.get('worker', [1])),
                             batch_size=64)
        for epoch in range(args.epochs):
            for batch in ds:           # list of decoded Examples
                ...
    """

    def __init__(self, path, shard_index=0, num_shards=1, batch_size=1,
                 decode=True, drop_remainder=False, shuffle_buffer=0, seed=0):
        if path.startswith("file://"):
            path = path[len("file://"):]
        self.files = shard_files(path, shard_index, num_shards)
        if not self.files:
            raise FileNotFoundError(
                "no TFRecord files for shard {}/{} under {}".format(
                    shard_index, num_shards, path))
        self.batch_size = batch_size
        self.decode = decode
        self.drop_remainder = drop_remainder
        self.shuffle_buffer = shuffle_buffer
        self.seed = seed
        self._epoch = 0

    def _records(self):
        files = list(self.files)
        if self.shuffle_buffer:
            import random
            rng = random.Random(hash((self.seed, self._epoch)))
            rng.shuffle(files)
            buf = []
            for ex in tfrecord_examples(files, self.decode):
                buf.append(ex)
                if len(buf) >= self.shuffle_buffer:
                    yield buf.pop(rng.randrange(len(buf)))
            while buf:
                yield buf.pop(rng.randrange(len(buf)))
        else:
            yield from tfrecord_examples(files, self.decode)

    def __iter__(self):
        self._epoch += 1  # new shuffle order per epoch
        return batched(self._records(), self.batch_size, self.drop_remainder)

    def count(self):
        return sum(1 for _ in tfrecord_examples(self.files, decode=False))
