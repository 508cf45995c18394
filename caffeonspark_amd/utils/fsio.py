"""Filesystem-URI IO: local paths plus remote schemes through fsspec.

The reference reads LMDB/seqfiles/DataFrames from HDFS and uploads
snapshots there (FSUtils.scala:49-89: GenModelOrState copies the local
snapshot up; GetLocalFileName pulls an HDFS snapshot to cwd for resume;
LmdbRDD distributes the db via sc.addFile, LmdbRDD.scala:252-258).

Same shape here: `ensure_local(uri)` materialises a remote source in a
local cache dir before the (mmap-based) readers touch it, `copy_to_uri`
uploads a produced artifact.  Remote access goes through fsspec, so
`hdfs://` (via pyarrow libhdfs when present) and `webhdfs://` (pure
requests) both work where those drivers exist; plain paths and `file:`
URIs never touch fsspec.
"""

from __future__ import annotations

import os
import shutil
import tempfile

_LOCAL_SCHEMES = ("file",)


def split_scheme(uri: str):
    """(scheme, rest) — scheme '' for plain paths. Windows-style drive
    letters don't occur here; anything before '://' is the scheme."""
    if "://" in uri:
        scheme, rest = uri.split("://", 1)
        return scheme.lower(), rest
    if uri.startswith("file:"):
        return "file", uri[5:]
    return "", uri


def is_remote(uri: str) -> bool:
    scheme, _ = split_scheme(uri)
    return scheme not in ("",) + _LOCAL_SCHEMES


def _fs(scheme: str):
    import fsspec
    return fsspec.filesystem(scheme)


def ensure_local(uri: str, cache_dir: str = None) -> str:
    """Local path for `uri`: plain/file paths pass through; remote URIs
    are downloaded once into cache_dir (reference FSUtils.GetLocalFileName
    pulls the HDFS snapshot next to the executor)."""
    scheme, rest = split_scheme(uri)
    if scheme in ("",) + _LOCAL_SCHEMES:
        return rest if scheme else uri
    cache_dir = cache_dir or os.path.join(tempfile.gettempdir(),
                                          "cosamd_fscache")
    os.makedirs(cache_dir, exist_ok=True)
    local = os.path.join(cache_dir, rest.strip("/").replace("/", "_"))
    fs = _fs(scheme)
    if fs.isdir(uri):
        if not os.path.isdir(local):
            fs.get(uri, local, recursive=True)
    elif not os.path.exists(local):
        fs.get(uri, local)
    return local


def copy_to_uri(local_path: str, uri: str) -> None:
    """Upload a local artifact to `uri` (reference FSUtils.GenModelOrState:
    snapshot then move/upload by filename convention)."""
    scheme, rest = split_scheme(uri)
    if scheme in ("",) + _LOCAL_SCHEMES:
        dest = rest if scheme else uri
        os.makedirs(os.path.dirname(dest) or ".", exist_ok=True)
        shutil.copyfile(local_path, dest)
        return
    fs = _fs(scheme)
    parent = uri.rsplit("/", 1)[0]
    try:
        fs.makedirs(parent, exist_ok=True)
    except Exception:
        pass
    fs.put(local_path, uri)


def open_uri(uri: str, mode: str = "rb"):
    scheme, rest = split_scheme(uri)
    if scheme in ("",) + _LOCAL_SCHEMES:
        return open(rest if scheme else uri, mode)
    import fsspec
    return fsspec.open(uri, mode).open()
