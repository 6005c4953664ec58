"""Legacy packaging facade (parity with reference ``tf_yarn/packaging.py``,
itself a thin re-export after the logic moved to cluster_pack).

On one node there is no environment shipping: the spawner puts the package
root on each task's PYTHONPATH (see ``client._task_env``).  These helpers
keep the old call sites working."""

from __future__ import annotations

import logging
import os
import zipfile

logger = logging.getLogger(__name__)


def zip_path(py_dir: str, include_base_name: bool = True,
             tmp_dir: str = "/tmp") -> str:
    """Zip a directory (reference ``packaging.py:23-36``)."""
    base = os.path.basename(py_dir.rstrip("/"))
    out = os.path.join(tmp_dir, f"{base}.zip")
    with zipfile.ZipFile(out, "w", zipfile.ZIP_DEFLATED) as zf:
        for root, _, files in os.walk(py_dir):
            for f in files:
                full = os.path.join(root, f)
                rel = os.path.relpath(full, py_dir)
                if include_base_name:
                    rel = os.path.join(base, rel)
                zf.write(full, rel)
    return out


def upload_env(package_path: str = None, *args, **kwargs):
    """No-op on a single node (reference shipped the env to HDFS)."""
    logger.info("upload_env: single-node build, nothing to upload")
    return package_path, None


# Backwards-compatible alias (reference `upload_env_to_hdfs`).
upload_env_to_hdfs = upload_env
