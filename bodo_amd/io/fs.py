"""Filesystem abstraction (reference role: bodo/libs/_fs_io.cpp +
fs_io.py Arrow-FS routing): paths with a scheme (s3://, gs://, hdfs://,
file://) resolve through pyarrow.fs.FileSystem.from_uri; bare paths stay on
the fast local-OS path (readinto-pinned prefetch).  The offline image has no
object store to talk to, but every reader/writer routes through here so an
S3 deployment is a URI away."""

from __future__ import annotations

from typing import Optional, Tuple

import pyarrow as pa
import pyarrow.fs as pafs


def resolve(path: str) -> Tuple[Optional[pafs.FileSystem], str]:
    """Returns (filesystem, fs-relative path); filesystem None = local OS
    path (callers may use plain open())."""
    if "://" in path:
        fs, p = pafs.FileSystem.from_uri(path)
        if isinstance(fs, pafs.LocalFileSystem):
            return None, p
        return fs, p
    return None, path


def is_remote(path: str) -> bool:
    fs, _ = resolve(path)
    return fs is not None


def open_input(path: str):
    fs, p = resolve(path)
    if fs is None:
        return open(p, "rb")
    return fs.open_input_file(p)


def open_output(path: str):
    fs, p = resolve(path)
    if fs is None:
        return open(p, "wb")
    return fs.open_output_stream(p)


def makedirs(path: str) -> None:
    fs, p = resolve(path)
    if fs is None:
        import os

        os.makedirs(p, exist_ok=True)
    else:
        fs.create_dir(p, recursive=True)


def list_files(path: str, suffix: str = "") -> list:
    fs, p = resolve(path)
    if fs is None:
        import glob
        import os

        if os.path.isdir(p):
            return sorted(glob.glob(os.path.join(p, f"*{suffix}")))
        return [p]
    info = fs.get_file_info(p)
    if info.type == pafs.FileType.Directory:
        sel = pafs.FileSelector(p, recursive=False)
        return sorted(f.path for f in fs.get_file_info(sel)
                      if f.is_file and f.path.endswith(suffix))
    return [p]
