"""Filesystem Iceberg table subset: versioned snapshot metadata with
transactional commits and time travel over parquet data files.

Layout of a table directory::

    <table>/metadata/version-hint.text      latest metadata version number
    <table>/metadata/v<N>.metadata.json     schema + snapshot log
    <table>/metadata/snap-<id>.json         full data-file list of a snapshot
    <table>/data/part-<snap>-<rank>.parquet data files

Commit protocol (the reference's resume unit for ETL jobs —
bodo/io/iceberg/write.py commits metadata only after every rank's files
land): all ranks write their data files, file infos are gathered, and
rank 0 publishes the snapshot + metadata json and finally atomically
replaces version-hint.text; readers only ever follow version-hint, so a
torn write is invisible.

Scope note: metadata is JSON, not Avro manifests, and there is no REST /
Glue / Snowflake catalog — the offline image has neither pyiceberg nor an
Avro codec.  The layout mirrors Iceberg's version-hint/metadata/snapshot
structure so a catalog adapter can be layered on without touching the
engine.  (Reference surface: bodo/pandas/base.py read_iceberg:313,
frame.py to_iceberg, bodo/io/iceberg/.)
"""

from __future__ import annotations

import json
import os
import tempfile
import time
from typing import List, Optional

SNAP_PREFIX = "iceberg-snapshot="


def _meta_dir(path: str) -> str:
    return os.path.join(path, "metadata")


def is_iceberg_dir(path: str) -> bool:
    return os.path.isfile(os.path.join(_meta_dir(path), "version-hint.text"))


def split_snapshot(path: str):
    """'dir@iceberg-snapshot=N' -> (dir, N); plain path -> (path, None)."""
    if "@" + SNAP_PREFIX in path:
        base, _, snap = path.rpartition("@" + SNAP_PREFIX)
        return base, int(snap)
    return path, None


def _current_version(path: str) -> int:
    with open(os.path.join(_meta_dir(path), "version-hint.text")) as f:
        return int(f.read().strip())


def read_metadata(path: str) -> dict:
    v = _current_version(path)
    with open(os.path.join(_meta_dir(path), f"v{v}.metadata.json")) as f:
        return json.load(f)


def snapshots(path: str) -> List[dict]:
    """Snapshot log of a table (id, timestamp-ms, operation, file count)."""
    return read_metadata(path)["snapshots"]


def data_files(path: str, snapshot_id: Optional[int] = None) -> List[str]:
    meta = read_metadata(path)
    snaps = meta["snapshots"]
    if not snaps:
        return []
    if snapshot_id is None:
        snap = next(s for s in snaps
                    if s["snapshot-id"] == meta["current-snapshot-id"])
    else:
        match = [s for s in snaps if s["snapshot-id"] == snapshot_id]
        if not match:
            raise ValueError(
                f"snapshot {snapshot_id} not found in {path} "
                f"(have {[s['snapshot-id'] for s in snaps]})")
        snap = match[0]
    with open(os.path.join(path, snap["manifest"])) as f:
        manifest = json.load(f)
    return [os.path.join(path, e["path"]) for e in manifest["files"]]


def resolve_scan_path(path: str) -> Optional[List[str]]:
    """Iceberg dir (optionally with @iceberg-snapshot=N) -> data file list;
    None when the path is not an iceberg table."""
    base, snap = split_snapshot(path)
    if not is_iceberg_dir(base):
        return None
    return data_files(base, snap)


def _atomic_write(path: str, text: str) -> None:
    d = os.path.dirname(path)
    fd, tmp = tempfile.mkstemp(dir=d)
    with os.fdopen(fd, "w") as f:
        f.write(text)
    os.replace(tmp, path)


def write_iceberg(tbl, path: str, mode: str, ctx) -> None:
    """Distributed transactional write.  mode: 'create' (error if the table
    exists), 'replace', or 'append'."""
    import pyarrow.parquet as pq

    from ..parallel import comm

    exists = is_iceberg_dir(path)
    if mode == "create" and exists:
        raise FileExistsError(f"iceberg table already exists: {path}")
    if mode == "append" and not exists:
        mode = "create"
    if mode not in ("create", "replace", "append"):
        raise ValueError(f"unknown iceberg write mode {mode!r}")

    os.makedirs(os.path.join(path, "data"), exist_ok=True)
    os.makedirs(_meta_dir(path), exist_ok=True)
    snap_id = int(time.time() * 1000)
    at = tbl.to_device("cpu").to_arrow()
    rel = os.path.join("data", f"part-{snap_id}-{ctx.rank:05d}.parquet")
    full = os.path.join(path, rel)
    pq.write_table(at, full)
    info = {"path": rel, "record_count": at.num_rows,
            "file_size_in_bytes": os.path.getsize(full)}
    infos = comm.allgather_obj(info) if ctx.world > 1 else [info]
    if ctx.rank == 0:
        files = list(infos)
        old_snaps: List[dict] = []
        version = 0
        if exists:
            meta = read_metadata(path)
            version = _current_version(path)
            old_snaps = meta["snapshots"]
            if mode == "append" and old_snaps:
                cur = next(s for s in old_snaps
                           if s["snapshot-id"] == meta["current-snapshot-id"])
                with open(os.path.join(path, cur["manifest"])) as f:
                    files = json.load(f)["files"] + files
        manifest_rel = os.path.join("metadata", f"snap-{snap_id}.json")
        _atomic_write(os.path.join(path, manifest_rel),
                      json.dumps({"files": files}))
        snap = {"snapshot-id": snap_id,
                "timestamp-ms": snap_id,
                "operation": mode,
                "manifest": manifest_rel,
                "total-records": sum(e["record_count"] for e in files),
                "total-data-files": len(files)}
        meta_out = {
            "format-version": 2,
            "table-uuid": f"bodo-amd-{snap_id}",
            "location": os.path.abspath(path),
            "schema": {"fields": [
                {"name": f.name, "type": str(f.type)} for f in at.schema]},
            "current-snapshot-id": snap_id,
            "snapshots": (old_snaps if mode == "append" else []) + [snap],
        }
        _atomic_write(
            os.path.join(_meta_dir(path), f"v{version + 1}.metadata.json"),
            json.dumps(meta_out, indent=1))
        # the commit point: readers follow version-hint only
        _atomic_write(os.path.join(_meta_dir(path), "version-hint.text"),
                      str(version + 1))
    if ctx.world > 1:
        comm.barrier()


def merge_into(path: str, source, on, when_matched: str = "update",
               when_not_matched: str = "insert"):
    """Copy-on-write MERGE INTO for filesystem Iceberg tables (reference:
    bodo/io/iceberg/merge_into.py).  Upsert semantics composed from the
    engine's distributed joins: surviving target rows = anti-join on the
    merge keys; matched updates take the source row wholesale; unmatched
    source rows insert.  The rewritten table commits as a new transactional
    snapshot (mode='replace'), so readers see either the old or the new
    snapshot, never a mix."""
    import bodo_amd.pandas as bpd

    from ..pandas.frame import BodoDataFrame
    from ..plan import nodes as pn

    keys = [on] if isinstance(on, str) else list(on)
    if when_matched not in ("update", "delete"):
        raise ValueError(f"when_matched={when_matched!r}")
    if when_not_matched not in ("insert", "ignore"):
        raise ValueError(f"when_not_matched={when_not_matched!r}")
    tgt = bpd.read_iceberg(path)
    cols = list(tgt.columns)
    if not isinstance(source, BodoDataFrame):
        source = bpd.from_pandas(source)
    missing = [c for c in cols if c not in list(source.columns)]
    if missing:
        raise ValueError(f"source is missing target columns {missing}")
    src = source[cols]

    survivors = BodoDataFrame(
        pn.Join(tgt._plan, src._plan, tuple(keys), tuple(keys), "anti"),
        cols)
    parts = [survivors]
    if when_matched == "update" and when_not_matched == "insert":
        parts.append(src)
    elif when_matched == "update":
        parts.append(BodoDataFrame(
            pn.Join(src._plan, tgt._plan, tuple(keys), tuple(keys), "semi"),
            cols))
    elif when_not_matched == "insert":
        parts.append(BodoDataFrame(
            pn.Join(src._plan, tgt._plan, tuple(keys), tuple(keys), "anti"),
            cols))
    out = parts[0] if len(parts) == 1 else bpd.concat(parts)
    out.to_iceberg(path, mode="replace")
