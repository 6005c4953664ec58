"""Filesystem abstraction with the ``resolve_filesystem_and_path`` seam.

The reference reaches HDFS through ``cluster_pack.filesystem``
(``model_ckpt.py:19,46,65``, ``parquet_dataset.py:21``); here the same
URI -> (fs, path) seam resolves to a local/posix implementation (and pyarrow
filesystems for URIs pyarrow knows), so checkpoint-layout code is identical
to the reference's shape without a Hadoop dependency.
"""

from __future__ import annotations

import os
import shutil
from typing import List, Tuple


class LocalFs:
    """Minimal filesystem object with the method set the reference's
    checkpoint/dataset code uses (exists/put/get/ls/mkdir/open/rm)."""

    def exists(self, path: str) -> bool:
        return os.path.exists(path)

    def ls(self, path: str) -> List[str]:
        if not os.path.isdir(path):
            return []
        return sorted(os.path.join(path, p) for p in os.listdir(path))

    def mkdir(self, path: str) -> None:
        os.makedirs(path, exist_ok=True)

    def put(self, local_src: str, dst: str) -> None:
        os.makedirs(os.path.dirname(dst) or ".", exist_ok=True)
        if os.path.isdir(local_src):
            shutil.copytree(local_src, dst, dirs_exist_ok=True)
        else:
            shutil.copy2(local_src, dst)

    def get(self, src: str, local_dst: str) -> None:
        os.makedirs(os.path.dirname(local_dst) or ".", exist_ok=True)
        shutil.copy2(src, local_dst)

    def rm(self, path: str, recursive: bool = False) -> None:
        if os.path.isdir(path):
            if recursive:
                shutil.rmtree(path, ignore_errors=True)
            else:
                os.rmdir(path)
        elif os.path.exists(path):
            os.remove(path)

    def open(self, path: str, mode: str = "rb"):
        if "w" in mode or "a" in mode:
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        return open(path, mode)


def resolve_filesystem_and_path(uri: str) -> Tuple[LocalFs, str]:
    """URI -> (filesystem, path).  ``file://`` and bare paths resolve to the
    local filesystem; other schemes raise (no remote FS on one node)."""
    if uri.startswith("file://"):
        return LocalFs(), uri[len("file://"):]
    if "://" in uri:
        scheme = uri.split("://", 1)[0]
        raise ValueError(
            f"unsupported filesystem scheme {scheme!r} "
            "(single-node build supports local paths and file://)")
    return LocalFs(), uri
