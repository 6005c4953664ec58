"""Optional MLflow facade (parity with reference ``tf_yarn/mlflow.py``).

Every call silently no-ops when mlflow is not importable or no tracking URI
is configured; connection errors are swallowed (reference ``mlflow.py:20-57``).
The ``TF_YARN_USE_MLFLOW`` env kill-switch is honored (``mlflow.py:28``).
"""

from __future__ import annotations

import functools
import logging
import os
import tempfile
from typing import Any, Dict, Optional

logger = logging.getLogger(__name__)

try:
    import mlflow as _mlflow  # type: ignore
    from mlflow.exceptions import MlflowException  # type: ignore
    _MLFLOW_IMPORTABLE = True
except ImportError:
    _mlflow = None
    MlflowException = Exception  # type: ignore
    _MLFLOW_IMPORTABLE = False


def _detect_mlflow() -> bool:
    if os.environ.get("TF_YARN_USE_MLFLOW", "").lower() in ("false", "0"):
        return False
    if not _MLFLOW_IMPORTABLE:
        return False
    try:
        return bool(_mlflow.get_tracking_uri())
    except Exception:
        return False


use_mlflow: bool = _detect_mlflow()


def optional_mlflow(return_default: Any = None):
    """Decorator: run the wrapped call only when mlflow is usable, and
    swallow connection errors (reference ``mlflow.py:57-70``)."""
    def decorator(f):
        @functools.wraps(f)
        def wrapper(*args, **kwargs):
            if use_mlflow:
                try:
                    return f(*args, **kwargs)
                except (ConnectionError, MlflowException) as e:
                    logger.warning("mlflow call failed: %s", e)
            return return_default
        return wrapper
    return decorator


@optional_mlflow()
def active_run_id() -> str:
    run = _mlflow.active_run()
    if run is None:
        run = _mlflow.start_run()
    return run.info.run_id


@optional_mlflow()
def get_tracking_uri() -> str:
    return _mlflow.get_tracking_uri()


@optional_mlflow()
def set_tag(key: str, value: Any) -> None:
    _mlflow.set_tag(key, value)


@optional_mlflow()
def set_tags(tags: Dict[str, Any]) -> None:
    _mlflow.set_tags(tags)


@optional_mlflow()
def log_param(key: str, value: Any) -> None:
    _mlflow.log_param(key, value)


@optional_mlflow()
def log_params(params: Dict[str, Any]) -> None:
    _mlflow.log_params(params)


@optional_mlflow()
def log_metric(key: str, value: float, step: Optional[int] = None) -> None:
    _mlflow.log_metric(key, value, step)


@optional_mlflow()
def log_metrics(metrics: Dict[str, float],
                step: Optional[int] = None) -> None:
    _mlflow.log_metrics(metrics, step)


def format_key(key: str) -> str:
    """Replace mlflow-hostile chars (reference ``mlflow.py:126-131``)."""
    return key.replace(":", "_").replace("/", "_") if key else ""


@optional_mlflow()
def save_text_to_mlflow(content: str, filename: str) -> None:
    """Reference ``mlflow.py:134-144``."""
    with tempfile.TemporaryDirectory() as tmpdir:
        path = os.path.join(tmpdir, filename)
        with open(path, "w") as fd:
            fd.write(content)
        _mlflow.log_artifact(path)
