"""EvaluatorMetricsLogger threshold/dedup behavior, table-driven over KV
snapshots (reference tests/test_evaluator_metrics.py:14-75)."""

import logging

import pytest

from tf_yarn_amd.evaluator_metrics import (MONITORED_METRICS,
                                           EvaluatorMetricsLogger)


def _fill(kv_client, evaluator, values):
    for metric, value in values.items():
        kv_client.put(f"{evaluator}/{metric}", str(value).encode())


def test_logs_all_metrics_once(kv_client, caplog):
    _fill(kv_client, "evaluator:0", {
        "awake_time_ratio": 0.5, "eval_step_mean_duration": 0.1,
        "last_training_step": 100, "nb_eval_steps": 10})
    logger = EvaluatorMetricsLogger(["evaluator:0"], kv_client)
    with caplog.at_level(logging.INFO):
        logger.log()
    messages = [r.message for r in caplog.records]
    assert any("Awake/idle ratio" in m for m in messages)
    assert any("100" in m for m in messages)


def test_dedups_repeated_values(kv_client, caplog):
    _fill(kv_client, "evaluator:0", {"nb_eval_steps": 5})
    logger = EvaluatorMetricsLogger(["evaluator:0"], kv_client)
    with caplog.at_level(logging.INFO):
        logger.log()
        n_first = len(caplog.records)
        logger.log()  # same value: no new log lines
    assert len(caplog.records) == n_first
    _fill(kv_client, "evaluator:0", {"nb_eval_steps": 6})
    with caplog.at_level(logging.INFO):
        logger.log()
    assert len(caplog.records) > n_first


@pytest.mark.parametrize("value,lo,hi,should_log", [
    (0.5, 0.2, 0.8, True),
    (0.1, 0.2, 0.8, False),
    (0.9, 0.2, 0.8, False),
    (0.9, 0.2, None, True),
    (0.1, None, 0.8, True),
])
def test_thresholds(kv_client, caplog, value, lo, hi, should_log):
    _fill(kv_client, "evaluator:0", {"awake_time_ratio": value})
    logger = EvaluatorMetricsLogger(
        ["evaluator:0"], kv_client,
        log_thresholds={"awake_time_ratio": [lo, hi]})
    with caplog.at_level(logging.INFO):
        logger.log()
    logged = any("Awake/idle ratio" in r.message for r in caplog.records)
    assert logged is should_log


def test_missing_keys_are_skipped(kv_client, caplog):
    logger = EvaluatorMetricsLogger(["evaluator:0"], kv_client)
    with caplog.at_level(logging.INFO):
        logger.log()
    assert not caplog.records


def test_monitored_metric_names_parity():
    """Reference evaluator_metrics.py:12-17."""
    assert set(MONITORED_METRICS) == {
        "awake_time_ratio", "eval_step_mean_duration",
        "last_training_step", "nb_eval_steps"}
