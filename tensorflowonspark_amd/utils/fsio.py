"""Scheme-aware file I/O for TFRecord paths (hdfs://, s3a://, file://, ...).

The reference reached HDFS through the tensorflow-hadoop Input/OutputFormats
(reference ``dfutil.py:39-41,63-65``) and expanded the libhdfs classpath for
TF's HDFS reader (reference ``TFSparkNode.py:284-290``). Here the same
capability is provided through ``fsspec`` (with ``pyarrow.fs`` as the hdfs
driver fsspec delegates to): any path carrying a URI scheme is opened through
the matching fsspec filesystem, plain paths stay on the fast builtin ``open``.

``TFNode.hdfs_path`` produces the schemed URIs; this module makes them
openable by ``tfrecord.TFRecordWriter`` / ``tfrecord_iterator`` /
``dataset.shard_files`` / ``dfutil.saveAsTFRecords``.
"""

import logging
import os
import re

logger = logging.getLogger(__name__)

# scheme detection: "hdfs://...", "s3a://...", "memory://..." — but NOT
# windows drive letters ("C:/...") or bare posix paths
_SCHEME_RE = re.compile(r"^([A-Za-z][A-Za-z0-9+.-]+)://")


def get_scheme(path):
    """URI scheme of ``path`` or None for plain/relative posix paths."""
    m = _SCHEME_RE.match(str(path))
    return m.group(1) if m else None


def _fs_for(path):
    """(fsspec_filesystem, path_inside_fs) for a schemed path."""
    import fsspec
    fs, fspath = fsspec.core.url_to_fs(path)
    return fs, fspath


def fs_open(path, mode="rb"):
    """Open ``path`` for reading/writing; fsspec for schemed paths."""
    scheme = get_scheme(path)
    if scheme is None or scheme == "file":
        local = path[len("file://"):] if scheme == "file" else path
        if "w" in mode or "a" in mode:
            d = os.path.dirname(os.path.abspath(local))
            if d:
                os.makedirs(d, exist_ok=True)
        return open(local, mode)
    fs, fspath = _fs_for(path)
    if "w" in mode or "a" in mode:
        parent = fspath.rsplit("/", 1)[0]
        if parent:
            try:
                fs.makedirs(parent, exist_ok=True)
            except Exception:  # some filesystems have no real directories
                pass
    return fs.open(fspath, mode)


def fs_exists(path):
    scheme = get_scheme(path)
    if scheme is None or scheme == "file":
        local = path[len("file://"):] if scheme == "file" else path
        return os.path.exists(local)
    fs, fspath = _fs_for(path)
    return fs.exists(fspath)


def fs_isdir(path):
    scheme = get_scheme(path)
    if scheme is None or scheme == "file":
        local = path[len("file://"):] if scheme == "file" else path
        return os.path.isdir(local)
    fs, fspath = _fs_for(path)
    return fs.isdir(fspath)


def fs_makedirs(path):
    scheme = get_scheme(path)
    if scheme is None or scheme == "file":
        local = path[len("file://"):] if scheme == "file" else path
        os.makedirs(local, exist_ok=True)
        return
    fs, fspath = _fs_for(path)
    try:
        fs.makedirs(fspath, exist_ok=True)
    except Exception:
        pass


def fs_glob(pattern):
    """Glob a possibly-schemed pattern; returns paths with the scheme kept
    (so results round-trip back into :func:`fs_open`)."""
    scheme = get_scheme(pattern)
    if scheme is None:
        import glob as _glob
        return sorted(_glob.glob(pattern))
    if scheme == "file":
        import glob as _glob
        return sorted("file://" + p for p in _glob.glob(pattern[len("file://"):]))
    fs, fspath = _fs_for(pattern)
    prefix = "{}://".format(scheme)
    out = []
    for p in fs.glob(fspath):
        p = str(p)
        out.append(p if _SCHEME_RE.match(p) else prefix + p.lstrip("/")
                   if scheme in ("memory",) else prefix + p)
    return sorted(out)


def fs_listfiles(directory):
    """Sorted data files directly under ``directory`` (schemed or plain),
    skipping hidden/_SUCCESS-style entries; scheme kept on results."""
    scheme = get_scheme(directory)
    if scheme is None or scheme == "file":
        local = directory[len("file://"):] if scheme == "file" else directory
        names = sorted(
            os.path.join(local, f) for f in os.listdir(local)
            if os.path.isfile(os.path.join(local, f))
            and not f.startswith(("_", ".")))
        if scheme == "file":
            return ["file://" + p for p in names]
        return names
    fs, fspath = _fs_for(directory)
    prefix = "{}://".format(scheme)
    out = []
    for p in fs.ls(fspath, detail=True):
        if p.get("type") == "file":
            name = str(p["name"])
            base = name.rsplit("/", 1)[-1]
            if base.startswith(("_", ".")):
                continue
            out.append(name if _SCHEME_RE.match(name)
                       else prefix + name.lstrip("/") if scheme in ("memory",)
                       else prefix + name)
    return sorted(out)
