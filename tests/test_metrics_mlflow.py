"""mlflow facade no-op semantics + OneShotMetricsLogger one-shot polling
(reference tf_yarn/mlflow.py:20-70, tf_yarn/metrics.py:41-59)."""

import logging

from tf_yarn_amd import metrics as metrics_mod
from tf_yarn_amd import mlflow as mlflow_shim
from tf_yarn_amd.metrics import Metrics, OneShotMetricsLogger


def test_mlflow_calls_are_noops_without_tracking():
    # no tracking URI configured in tests: every facade call must be a
    # silent no-op returning the decorator default
    assert mlflow_shim.use_mlflow is False
    assert mlflow_shim.active_run_id() is None
    assert mlflow_shim.log_metric("k", 1.0) is None
    assert mlflow_shim.set_tag("k", "v") is None
    assert mlflow_shim.save_text_to_mlflow("text", "name") is None


def test_format_key_replaces_separators():
    assert mlflow_shim.format_key("worker:0/stop") == "worker_0_stop"
    assert mlflow_shim.format_key("") == ""


def test_optional_mlflow_decorator_returns_default():
    @mlflow_shim.optional_mlflow(return_default=42)
    def would_blow_up():
        raise ConnectionError("unreachable")
    assert would_blow_up() == 42


def test_one_shot_metrics_logger(kv_client, caplog):
    logger = OneShotMetricsLogger(
        kv_client,
        [("tensorboard:0/url", "Tensorboard is listening at"),
         ("worker:0/logs", "worker logs")])
    with caplog.at_level(logging.INFO, logger=metrics_mod.__name__):
        logger.log()  # nothing published yet
        assert len(logger.events) == 2
        kv_client.put("tensorboard:0/url", b"http://host:6006")
        logger.log()
        assert len(logger.events) == 1  # url logged exactly once
        logger.log()
        assert len(logger.events) == 1  # ...and not re-logged
    assert any("http://host:6006" in m for m in caplog.messages)


def test_metrics_log_mlflow_is_noop_without_tracking():
    m = Metrics(total_training_duration=1.0, total_eval_duration=None,
                container_duration={"chief:0": 1.0},
                train_eval_time_per_node={"chief:0": 0.9})
    m.log_mlflow(n_try=0)  # must not raise
