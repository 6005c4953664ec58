"""Lifecycle event vocabulary over the control-plane KV store.

Exact parity with the reference's ``tf_yarn/event.py``: key scheme
``"{type}:{id}/{stage}"`` with stages ``init`` (sock addr), ``start``,
``stop`` (formatted exception or ""), ``logs`` (URL/path), ``url``, and the
four timer keys.  Barriers, master election and exception propagation are all
built from ``broadcast`` + ``wait`` (reference ``event.py:13-85``).
"""

from __future__ import annotations

import logging
import time
import traceback
from typing import Optional

from tf_yarn_amd.kv import KVClient

logger = logging.getLogger(__name__)

CONTAINER_START_TIME = "container_start_time"
CONTAINER_STOP_TIME = "container_stop_time"
TRAIN_EVAL_START_TIME = "train_eval_start_time"
TRAIN_EVAL_STOP_TIME = "train_eval_stop_time"


def wait(client: KVClient, key: str,
         timeout: Optional[float] = None) -> str:
    """Block until *key* is set; return its value decoded
    (reference ``event.py:13-18``)."""
    logger.info("waiting for %s", key)
    value = client.wait(key, timeout=timeout).decode()
    logger.info("got %s = %r", key, value[:200])
    return value


def broadcast(client: KVClient, key: str, message: str = "") -> None:
    """Reference ``event.py:70-79``."""
    logger.info("broadcasting %s = %r", key, message[:200])
    client.put(key, message.encode())


def init_event(client: KVClient, task: str, sock_addr: str) -> None:
    broadcast(client, f"{task}/init", sock_addr)


def start_event(client: KVClient, task: str) -> None:
    broadcast(client, f"{task}/start")


def stop_event(client: KVClient, task: str,
               exception: Optional[BaseException]) -> None:
    broadcast(client, f"{task}/stop", maybe_format_exception(exception))


def logs_event(client: KVClient, task: str, logs: str) -> None:
    broadcast(client, f"{task}/logs", logs)


def url_event(client: KVClient, task: str, url: str) -> None:
    broadcast(client, f"{task}/url", url)


def broadcast_container_start_time(client: KVClient, task: str) -> None:
    broadcast(client, f"{task}/{CONTAINER_START_TIME}", str(time.time()))


def broadcast_container_stop_time(client: KVClient, task: str) -> None:
    broadcast(client, f"{task}/{CONTAINER_STOP_TIME}", str(time.time()))


def broadcast_train_eval_start_timer(client: KVClient, task: str) -> None:
    broadcast(client, f"{task}/{TRAIN_EVAL_START_TIME}", str(time.time()))


def broadcast_train_eval_stop_timer(client: KVClient, task: str) -> None:
    broadcast(client, f"{task}/{TRAIN_EVAL_STOP_TIME}", str(time.time()))


def maybe_format_exception(exception: Optional[BaseException]) -> str:
    """Reference ``event.py:82-85``."""
    if exception is None:
        return ""
    return "".join(traceback.format_exception(
        type(exception), exception, exception.__traceback__))
