"""Minimal TensorBoard event-file writer (no tensorboard dependency).

The reference spawned TensorBoard on chief:0 (reference ``TFSparkNode.py:292-
329``) and relied on user callbacks to write event files (``mnist_tf.py:57-
60``). Round 1 spawned the process but nothing wrote events (VERDICT r01
item 9). This module writes the event files directly: a TF event file is
TFRecord framing (masked CRC32-C, same codec as ``tfrecord.py``) around
``Event`` protobufs; scalars need only a handful of proto fields, hand-encoded
here with the same wire helpers the Example codec uses.

Event proto (tensorflow/core/util/event.proto):
    1: double wall_time      2: int64 step
    3: string file_version   5: Summary summary
Summary.Value: 1: string tag, 2: float simple_value.
"""

import os
import socket
import struct
import time

from .. import tfrecord


def _tag(field, wire):
    return tfrecord._varint((field << 3) | wire)


def _len_delim(field, payload):
    return _tag(field, 2) + tfrecord._varint(len(payload)) + payload


def _double(field, v):
    return _tag(field, 1) + struct.pack("<d", v)


def _float(field, v):
    return _tag(field, 5) + struct.pack("<f", v)


def _int64(field, v):
    return _tag(field, 0) + tfrecord._varint(v & 0xFFFFFFFFFFFFFFFF)


def encode_scalar_event(step, scalars, wall_time=None):
    """Event record bytes carrying {tag: float} simple values."""
    wall = time.time() if wall_time is None else wall_time
    values = b""
    for tag, v in scalars.items():
        val = _len_delim(1, tag.encode("utf-8")) + _float(2, float(v))
        values += _len_delim(1, val)
    return _double(1, wall) + _int64(2, int(step)) + _len_delim(5, values)


def decode_scalar_event(record):
    """Inverse of :func:`encode_scalar_event` (tests / inspection)."""
    pos, n = 0, len(record)
    wall, step, scalars = None, 0, {}
    while pos < n:
        key, pos = tfrecord._read_varint(record, pos)
        field, wire = key >> 3, key & 7
        if wire == 1:
            raw, pos = record[pos:pos + 8], pos + 8
            if field == 1:
                wall = struct.unpack("<d", raw)[0]
        elif wire == 0:
            v, pos = tfrecord._read_varint(record, pos)
            if field == 2:
                step = v
        elif wire == 2:
            ln, pos = tfrecord._read_varint(record, pos)
            payload, pos = record[pos:pos + ln], pos + ln
            if field == 5:  # summary: parse Value submessages
                p2 = 0
                while p2 < len(payload):
                    k2, p2 = tfrecord._read_varint(payload, p2)
                    l2, p2 = tfrecord._read_varint(payload, p2)
                    val, p2 = payload[p2:p2 + l2], p2 + l2
                    tag, sv, p3 = None, None, 0
                    while p3 < len(val):
                        k3, p3 = tfrecord._read_varint(val, p3)
                        f3, w3 = k3 >> 3, k3 & 7
                        if w3 == 2:
                            l3, p3 = tfrecord._read_varint(val, p3)
                            if f3 == 1:
                                tag = val[p3:p3 + l3].decode("utf-8")
                            p3 += l3
                        elif w3 == 5:
                            if f3 == 2:
                                sv = struct.unpack("<f", val[p3:p3 + 4])[0]
                            p3 += 4
                        elif w3 == 0:
                            _, p3 = tfrecord._read_varint(val, p3)
                        elif w3 == 1:
                            p3 += 8
                    if tag is not None and sv is not None:
                        scalars[tag] = sv
        elif wire == 5:
            pos += 4
    return wall, step, scalars


class SummaryWriter:
    """Append scalar events to ``logdir/events.out.tfevents.*``.

    API-compatible subset of torch.utils.tensorboard.SummaryWriter
    (add_scalar / add_scalars / flush / close); safe to construct on any
    rank but conventionally only the chief writes (compat.py semantics).
    """

    def __init__(self, logdir):
        os.makedirs(logdir, exist_ok=True)
        name = "events.out.tfevents.{}.{}".format(
            int(time.time()), socket.gethostname())
        self._w = tfrecord.TFRecordWriter(os.path.join(logdir, name))
        # version header record (what TensorBoard looks for first)
        self._w.write(_double(1, time.time()) +
                      _len_delim(3, b"brain.Event:2"))
        self._flush()

    def _flush(self):
        f = getattr(self._w, "_f", None)
        if f is not None and hasattr(f, "flush"):
            f.flush()

    def add_scalar(self, tag, value, step):
        self._w.write(encode_scalar_event(step, {tag: value}))

    def add_scalars(self, scalars, step):
        self._w.write(encode_scalar_event(step, scalars))

    def flush(self):
        self._flush()

    def close(self):
        self._w.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
