"""TFRecord <-> DataFrame conversion (parity: reference ``dfutil.py``).

Works on both real Spark DataFrames and the local :class:`LocalDataFrame`.
The TFRecord layer is the native codec in ``tfrecord.py`` (no tensorflow-hadoop
jar, no TF protobuf dependency).

Dtype mapping parity (reference ``dfutil.py:96-131``): float/double ->
FloatList, bool/ints -> Int64List, binary/string -> BytesList, arrays thereof.
Schema inference parity (reference ``dfutil.py:134-168``): int64 -> bigint,
float -> double, bytes -> string unless hinted in ``binary_features``;
multi-value features -> arrays.
"""

import glob
import logging
import os

from . import tfrecord

logger = logging.getLogger(__name__)

loadedDF = {}  # DataFrame provenance: df -> source dir (reference dfutil.py:18-26)


def isLoadedDF(df):
    """True if ``df`` was produced by :func:`loadTFRecords`."""
    return id(df) in loadedDF


def toTFExample(dtypes):
    """Returns an iterator closure converting DataFrame rows (in the column
    order of ``dtypes``) to serialized Example bytes."""
    cols = [name for name, _t in dtypes]
    types = [t for _n, t in dtypes]

    def _convert(iterator):
        for row in iterator:
            feats = {}
            for name, t, v in zip(cols, types, row):
                if v is None:
                    v = []
                if t in ("float", "double") or t.startswith("array<float>") \
                        or t.startswith("array<double>"):
                    feats[name] = [float(x) for x in (v if isinstance(v, (list, tuple)) else [v])]
                elif t in ("boolean", "tinyint", "smallint", "int", "bigint", "long") \
                        or t.startswith("array<int") or t.startswith("array<bigint") \
                        or t.startswith("array<long") or t.startswith("array<boolean"):
                    feats[name] = [int(x) for x in (v if isinstance(v, (list, tuple)) else [v])]
                elif t in ("binary", "string") or t.startswith("array<binary") \
                        or t.startswith("array<string"):
                    vals = v if isinstance(v, (list, tuple)) else [v]
                    feats[name] = [x if isinstance(x, (bytes, bytearray))
                                   else str(x).encode("utf-8") for x in vals]
                else:
                    raise TypeError("unsupported dtype {} for column {}".format(t, name))
            yield tfrecord.encode_example(feats)

    return _convert


def infer_schema(example, binary_features=None):
    """Infer (name, dtype) schema from one decoded Example
    ({name: (kind, values)})."""
    binary_features = set(binary_features or [])
    schema = []
    for name in sorted(example.keys()):
        kind, values = example[name]
        if kind == "int64":
            base = "bigint"
        elif kind == "float":
            base = "double"
        else:
            base = "binary" if name in binary_features else "string"
        if len(values) > 1:
            schema.append((name, "array<{}>".format(base)))
        else:
            schema.append((name, base))
    return schema


def fromTFExample(record, binary_features=None, schema=None):
    """Serialized Example -> row tuple (columns sorted by name)."""
    binary_features = set(binary_features or [])
    ex = tfrecord.decode_example(record)
    names = [n for n, _t in schema] if schema else sorted(ex.keys())
    row = []
    for name in names:
        if name not in ex:
            row.append(None)
            continue
        kind, values = ex[name]
        if kind == "bytes" and name not in binary_features:
            values = [v.decode("utf-8", errors="replace") for v in values]
        if len(values) == 0:
            row.append(None)
        elif len(values) == 1 and not (schema and
                                       dict(schema)[name].startswith("array")):
            row.append(values[0])
        else:
            row.append(list(values))
    return tuple(row)


def saveAsTFRecords(df, output_dir):
    """Save a DataFrame as TFRecord part files under ``output_dir``.

    Part files are written executor-side via the native codec. ``output_dir``
    may be a plain/shared posix path or a schemed URI (hdfs://, file://,
    s3a://, ...) opened through fsio/fsspec — the capability the reference got
    from the tensorflow-hadoop OutputFormat (``dfutil.py:39-41``)."""
    from .utils import fsio
    dtypes = df.dtypes
    convert = toTFExample(dtypes)
    fsio.fs_makedirs(output_dir)

    def _write(idx, iterator):
        path = output_dir.rstrip("/") + "/part-r-{:05d}".format(idx) \
            if fsio.get_scheme(output_dir) else \
            os.path.join(output_dir, "part-r-{:05d}".format(idx))
        n = 0
        with tfrecord.TFRecordWriter(path) as w:
            for rec in convert(iterator):
                w.write(rec)
                n += 1
        return [n]

    rdd = df.rdd
    if hasattr(rdd, "mapPartitionsWithIndex"):
        counts = rdd.mapPartitionsWithIndex(_write).collect()
        logger.info("wrote %s records to %s", sum(counts), output_dir)
    else:  # pragma: no cover
        single = output_dir.rstrip("/") + "/part-r-00000" \
            if fsio.get_scheme(output_dir) else \
            os.path.join(output_dir, "part-r-00000")
        with tfrecord.TFRecordWriter(single) as w:
            for rec in convert(iter(df.collect())):
                w.write(rec)


def loadTFRecords(sc, input_dir, binary_features=None, schema_hint=None):
    """Load TFRecord files under ``input_dir`` (plain path or schemed URI)
    as a DataFrame with inferred schema; records provenance in ``loadedDF``.

    ``schema_hint`` (a Spark-SQL simpleString like
    ``struct<label:bigint,image:array<double>>`` or a [(name, dtype)] list)
    overrides inference for the hinted fields and preserves their order —
    parity with the JVM ``DFUtil.loadTFRecords(schemaHint)``
    (reference ``DFUtil.scala:35-55,67-110``)."""
    from .utils import fsio
    if fsio.get_scheme(input_dir):
        files = fsio.fs_glob(input_dir.rstrip("/") + "/part-*") or \
            fsio.fs_listfiles(input_dir)
    else:
        files = sorted(f for f in glob.glob(os.path.join(input_dir, "part-*"))
                       if os.path.isfile(f))
        if not files:
            files = sorted(f for f in glob.glob(os.path.join(input_dir, "*"))
                           if os.path.isfile(f) and not os.path.basename(f).startswith("_"))
    if not files:
        raise FileNotFoundError("no TFRecord files under " + input_dir)

    first = next(tfrecord.tfrecord_iterator(files[0]))
    schema = infer_schema(tfrecord.decode_example(first), binary_features)
    if schema_hint is not None:
        hinted = parse_schema(schema_hint) if isinstance(schema_hint, str) \
            else list(schema_hint)
        rest = dict(schema)
        # hinted fields first (hinted order + dtypes), unhinted keep inference
        schema = hinted + [(n, t) for n, t in schema
                           if n not in dict(hinted)]
        schema = [(n, dict(hinted).get(n, rest.get(n))) for n, _ in schema]
        # a binary hint also switches the DECODER off utf-8 (like the JVM
        # loader's BinaryType handling, DFUtil.scala:119-184)
        binary_features = list(binary_features or []) + [
            n for n, t in hinted if t in ("binary", "array<binary>")]

    def _read(it):
        for path in it:
            for rec in tfrecord.tfrecord_iterator(path):
                yield fromTFExample(rec, binary_features, schema)

    rdd = sc.parallelize(files, min(len(files), sc.defaultParallelism)) \
        .mapPartitions(_read)
    if hasattr(sc, "createDataFrame"):
        # LocalSparkContext path: small data, driver-side assembly
        df = sc.createDataFrame(rdd.collect(), [n for n, _ in schema],
                                [t for _, t in schema])
    else:  # pragma: no cover - real pyspark path
        # keep the load distributed (no driver-side collect — the reference's
        # TFRecordFileInputFormat load was distributed too, dfutil.py:63-65)
        from pyspark.sql import SparkSession
        from pyspark.sql.types import (ArrayType, DoubleType, LongType,
                                       BinaryType, StringType, StructField,
                                       StructType)
        base = {"bigint": LongType(), "double": DoubleType(),
                "string": StringType(), "binary": BinaryType()}
        fields = []
        for n, t in schema:
            if t.startswith("array<"):
                fields.append(StructField(n, ArrayType(base[t[6:-1]])))
            else:
                fields.append(StructField(n, base[t]))
        spark = SparkSession.builder.getOrCreate()
        df = spark.createDataFrame(rdd, StructType(fields))
    loadedDF[id(df)] = input_dir
    return df


def parse_schema(simple_string):
    """Parse a Spark-SQL simpleString like ``struct<a:bigint,b:array<double>>``
    into a [(name, dtype)] schema (parity: reference SimpleTypeParser.scala:28
    — base types binary/boolean/int/long/bigint/float/double/string plus 1-D
    arrays)."""
    s = simple_string.strip()
    if s.startswith("struct<") and s.endswith(">"):
        s = s[len("struct<"):-1]
    fields = []
    depth = 0
    token = ""
    parts = []
    for ch in s:
        if ch == "<":
            depth += 1
        elif ch == ">":
            depth -= 1
        if ch == "," and depth == 0:
            parts.append(token)
            token = ""
        else:
            token += ch
    if token.strip():
        parts.append(token)
    valid = {"binary", "boolean", "int", "long", "bigint", "float", "double",
             "string", "tinyint", "smallint"}
    for part in parts:
        name, _, dtype = part.partition(":")
        name, dtype = name.strip(), dtype.strip()
        base = dtype[6:-1] if dtype.startswith("array<") and dtype.endswith(">") \
            else dtype
        if base not in valid:
            raise ValueError("unsupported type '{}' in schema".format(dtype))
        fields.append((name, dtype))
    return fields
