"""TFRecord file codec + tf.train.Example wire-format encode/decode.

Replaces the reference's vendored ``tensorflow-hadoop`` jar and TF protobuf
dependency (reference ``dfutil.py:29-81`` used
``org.tensorflow.hadoop.io.TFRecordFileInputFormat`` plus ``tf.train.Example``)
with a dependency-free implementation of both layers:

* **Framing**: ``[len:u64le][masked_crc32c(len):u32le][data][masked_crc32c(data)]``
  with CRC32-C (Castagnoli) and mask ``((crc>>15)|(crc<<17)) + 0xa282ead8``.
* **Example proto**: hand-written wire format for the fixed schema
  ``Example{Features{map<string,Feature>}}`` with
  ``Feature = BytesList | FloatList | Int64List`` — no protobuf runtime needed.

A C++ fast path (``csrc/tfrecord.cpp``) accelerates bulk scans; this module is
the reference implementation and the fallback.
"""

import os
import struct

# ---------------------------------------------------------------------------
# CRC32-C (Castagnoli, poly 0x82F63B78 reflected) with the TFRecord mask
# ---------------------------------------------------------------------------

_CRC_TABLE = []


def _build_table():
    poly = 0x82F63B78
    for i in range(256):
        c = i
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        _CRC_TABLE.append(c)


_build_table()


def crc32c(data, crc=0):
    crc ^= 0xFFFFFFFF
    tbl = _CRC_TABLE
    for b in data:
        crc = tbl[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def masked_crc(data):
    crc = crc32c(data)
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


# ---------------------------------------------------------------------------
# Record framing
# ---------------------------------------------------------------------------

class TFRecordWriter:
    def __init__(self, path):
        from .utils import fsio
        self._f = fsio.fs_open(path, "wb")  # scheme-aware (hdfs://, file://, ...)

    def write(self, record):
        length = struct.pack("<Q", len(record))
        self._f.write(length)
        self._f.write(struct.pack("<I", masked_crc(length)))
        self._f.write(record)
        self._f.write(struct.pack("<I", masked_crc(record)))

    def close(self):
        self._f.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


def _native_ext():
    try:
        from .ops import get_ext
        return get_ext(required=False)
    except Exception:
        return None


def tfrecord_iterator(path, verify=False):
    """Yield raw record bytes from a TFRecord file.

    Uses the C++ codec (HW CRC32-C, single mmap-style scan) when the extension
    is built; the pure-Python path below is the reference implementation."""
    from .utils import fsio
    scheme = fsio.get_scheme(path)
    local = path[len("file://"):] if scheme == "file" else path
    ext = _native_ext()
    if (scheme is None or scheme == "file") and ext is not None \
            and hasattr(ext, "tfrecord_read_file"):
        yield from ext.tfrecord_read_file(local, verify)
        return
    with fsio.fs_open(path, "rb") as f:
        while True:
            header = f.read(12)
            if len(header) < 12:
                return
            (length,) = struct.unpack("<Q", header[:8])
            if verify:
                (lcrc,) = struct.unpack("<I", header[8:12])
                if masked_crc(header[:8]) != lcrc:
                    raise IOError("corrupt TFRecord length crc in " + path)
            data = f.read(length)
            tail = f.read(4)
            if len(data) < length or len(tail) < 4:
                raise IOError("truncated TFRecord in " + path)
            if verify:
                (dcrc,) = struct.unpack("<I", tail)
                if masked_crc(data) != dcrc:
                    raise IOError("corrupt TFRecord data crc in " + path)
            yield data


# ---------------------------------------------------------------------------
# Protobuf wire helpers
# ---------------------------------------------------------------------------

def _varint(n):
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _read_varint(buf, pos):
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def _tag(field, wire):
    return _varint((field << 3) | wire)


def _len_delim(field, payload):
    return _tag(field, 2) + _varint(len(payload)) + payload


def _uint64(n):
    # int64 values are stored as plain varints (two's complement, 10 bytes if
    # negative) in Int64List — protobuf int64, not sint64/zigzag
    return n & 0xFFFFFFFFFFFFFFFF


# ---------------------------------------------------------------------------
# tf.train.Example encode
# ---------------------------------------------------------------------------

def _feature_bytes(values):
    payload = b"".join(_len_delim(1, v) for v in values)
    return _len_delim(1, payload)            # Feature.bytes_list = 1


def _feature_floats(values):
    packed = struct.pack("<{}f".format(len(values)), *values)
    payload = _len_delim(1, packed)           # FloatList.value packed
    return _len_delim(2, payload)             # Feature.float_list = 2


def _feature_int64s(values):
    packed = b"".join(_varint(_uint64(int(v))) for v in values)
    payload = _len_delim(1, packed)           # Int64List.value packed
    return _len_delim(3, payload)             # Feature.int64_list = 3


def encode_example(features):
    """features: {name: value}; value may be int/float/bool/str/bytes or a
    (numpy array or) list of those. Returns serialized Example bytes.

    Dtype mapping parity with reference ``dfutil.py:96-131``: float/double ->
    FloatList, bool/int/long -> Int64List, str/bytes -> BytesList.
    """
    import numpy as np
    body = b""
    for name, value in sorted(features.items()):
        if isinstance(value, np.ndarray):
            value = value.tolist()
        if not isinstance(value, (list, tuple)):
            value = [value]
        if len(value) == 0:
            feat = _len_delim(3, _len_delim(1, b""))
        elif isinstance(value[0], (bytes, bytearray)):
            feat = _feature_bytes([bytes(v) for v in value])
        elif isinstance(value[0], str):
            feat = _feature_bytes([v.encode("utf-8") for v in value])
        elif isinstance(value[0], (bool, int)) or isinstance(value[0], np.integer):
            feat = _feature_int64s(value)
        elif isinstance(value[0], float) or isinstance(value[0], np.floating):
            feat = _feature_floats(value)
        else:
            raise TypeError("unsupported feature type for {}: {}".format(
                name, type(value[0])))
        # map entry: key=1 (string), value=2 (Feature message);
        # feat is already the Feature message body (tag(kind)+len+payload)
        entry = _len_delim(1, name.encode("utf-8")) + _len_delim(2, feat)
        body += _len_delim(1, entry)              # Features.feature map entry
    example = _len_delim(1, body)                 # Example.features = 1
    return example


# ---------------------------------------------------------------------------
# tf.train.Example decode
# ---------------------------------------------------------------------------

def _parse_feature(buf):
    """Feature message -> (kind, [values]) with kind in {bytes,float,int64}."""
    pos = 0
    while pos < len(buf):
        key, pos = _read_varint(buf, pos)
        field, wire = key >> 3, key & 7
        assert wire == 2, "unexpected wire type in Feature"
        ln, pos = _read_varint(buf, pos)
        payload = buf[pos:pos + ln]
        pos += ln
        if field == 1:   # BytesList
            vals, p = [], 0
            while p < len(payload):
                k, p = _read_varint(payload, p)
                vlen, p = _read_varint(payload, p)
                vals.append(bytes(payload[p:p + vlen]))
                p += vlen
            return "bytes", vals
        if field == 2:   # FloatList
            vals, p = [], 0
            while p < len(payload):
                k, p = _read_varint(payload, p)
                f, w = k >> 3, k & 7
                if w == 2:  # packed
                    vlen, p = _read_varint(payload, p)
                    n = vlen // 4
                    vals.extend(struct.unpack("<{}f".format(n),
                                              payload[p:p + vlen]))
                    p += vlen
                else:       # unpacked fixed32
                    vals.append(struct.unpack("<f", payload[p:p + 4])[0])
                    p += 4
            return "float", vals
        if field == 3:   # Int64List
            vals, p = [], 0
            while p < len(payload):
                k, p = _read_varint(payload, p)
                f, w = k >> 3, k & 7
                if w == 2:  # packed varints
                    vlen, p = _read_varint(payload, p)
                    end = p + vlen
                    while p < end:
                        v, p = _read_varint(payload, p)
                        if v >= 1 << 63:
                            v -= 1 << 64
                        vals.append(v)
                else:
                    v, p = _read_varint(payload, p)
                    if v >= 1 << 63:
                        v -= 1 << 64
                    vals.append(v)
            return "int64", vals
    return "int64", []


def decode_example(record):
    """Serialized Example -> {name: (kind, [values])}."""
    out = {}
    pos = 0
    buf = memoryview(record)
    while pos < len(buf):
        key, pos = _read_varint(buf, pos)
        field, wire = key >> 3, key & 7
        if wire != 2:
            raise ValueError("unexpected wire type in Example")
        ln, pos = _read_varint(buf, pos)
        payload = bytes(buf[pos:pos + ln])
        pos += ln
        if field != 1:
            continue
        # Features message: repeated map entries (field 1)
        fpos = 0
        while fpos < len(payload):
            fkey, fpos = _read_varint(payload, fpos)
            ffield, fwire = fkey >> 3, fkey & 7
            flen, fpos = _read_varint(payload, fpos)
            entry = payload[fpos:fpos + flen]
            fpos += flen
            if ffield != 1:
                continue
            # map entry: key=1 string, value=2 Feature
            name, kindvals = None, None
            epos = 0
            while epos < len(entry):
                ekey, epos = _read_varint(entry, epos)
                efield = ekey >> 3
                elen, epos = _read_varint(entry, epos)
                evalue = entry[epos:epos + elen]
                epos += elen
                if efield == 1:
                    name = evalue.decode("utf-8")
                elif efield == 2:
                    kindvals = _parse_feature(evalue)
            if name is not None and kindvals is not None:
                out[name] = kindvals
    return out
