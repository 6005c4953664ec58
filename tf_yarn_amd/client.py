"""Core client: ``run_on_yarn`` as a single-node multi-process spawner.

This is the MI355X-native replacement for the reference's
``tf_yarn/client.py``: where the reference builds a skein ``ApplicationSpec``
(one YARN service per task role, ``client.py:179-263``) and polls the YARN
ApplicationMaster, this module maps the same role topology
(chief/worker/ps/evaluator/tensorboard) onto local processes pinned to the
node's 8 MI355X GPUs, coordinates them through the in-process
:class:`~tf_yarn_amd.kv.KVServer`, ships the ``experiment_fn`` closure by
cloudpickle through the store (reference ``client.py:281,536``), aggregates
per-task lifecycle events into :class:`~tf_yarn_amd.metrics.Metrics`
(reference ``client.py:633-739``) and retries the whole run ``nb_retries``
times (reference ``client.py:432-466``).
"""

from __future__ import annotations

import importlib
import json
import logging
import os
import signal
import subprocess
import sys
import threading
import time
import uuid
from functools import partial
from typing import Dict, List, NamedTuple, Optional, Tuple

import cloudpickle

from tf_yarn_amd import constants, event, mlflow
from tf_yarn_amd import _env as env_mod
from tf_yarn_amd._internal import iter_tasks
from tf_yarn_amd._task_commons import catchtime
from tf_yarn_amd.evaluator_metrics import EvaluatorMetricsLogger
from tf_yarn_amd.kv import KVClient, KVServer
from tf_yarn_amd.metrics import Metrics, OneShotMetricsLogger
from tf_yarn_amd.topologies import (ContainerTask, NodeLabel, TaskSpec,
                                    TaskSpecs, _check_general_topology,
                                    _check_ps_topology)

logger = logging.getLogger(__name__)

POLL_PERIOD_SECS = 0.5


class RunFailed(Exception):
    """Raised when the application ends with a failed task
    (reference ``client.py:89``)."""


class ContainerLogStatus(NamedTuple):
    """Reference ``client.py:61-86``."""
    log_paths: Dict[str, str] = {}
    statuses: Dict[str, str] = {}

    def by_container_id(self) -> Dict[str, Tuple[str, str]]:
        return {task: (self.log_paths.get(task, ""), status)
                for task, status in self.statuses.items()}


class TaskProcess(NamedTuple):
    task: ContainerTask
    process: subprocess.Popen
    log_path: str

    @property
    def key(self) -> str:
        return f"{self.task.type}:{self.task.id}"


class LocalCluster(NamedTuple):
    """The ``SkeinCluster`` equivalent (reference ``client.py:53-59``)."""
    server: KVServer
    client: KVClient
    app_id: str
    app_dir: str
    tasks: List[ContainerTask]
    processes: List[TaskProcess]
    event_listener: threading.Thread
    events: Dict[str, Dict[str, str]]


def get_safe_experiment_fn(full_fn_name: str, *args):
    """Import-by-name alternative to pickling the closure
    (reference ``client.py:472-495``)."""
    module_name, fn_name = full_fn_name.rsplit(".", 1)

    def _safe_exp_fn(*a):
        module = importlib.import_module(module_name)
        return getattr(module, fn_name)(*a)

    return partial(_safe_exp_fn, *args)


def _setup_cluster_spec(tasks: List[ContainerTask],
                        client: KVClient) -> None:
    """Write ``cluster_instances`` (excluding evaluator + tensorboard,
    reference ``client.py:170-176``)."""
    payload = [[t.type, t.id, t.nb_proc] for t in tasks
               if t.type not in ("evaluator", "tensorboard")]
    client.put(constants.KV_CLUSTER_INSTANCES,
               json.dumps(payload).encode())


def _allocate_gpus(tasks: List[ContainerTask],
                   task_specs: TaskSpecs) -> Dict[str, List[int]]:
    """Assign node GPUs round-robin to GPU-labelled training tasks.

    Replaces the reference's in-worker round-robin
    (``pytorch/tasks/worker.py:162-168``): the spawner owns the whole node so
    assignment is static — contiguous GPU ids per container keep each
    container's processes on xGMI-adjacent devices.
    """
    # Clamp the node model (NODE_GPU_COUNT) to what the machine really
    # has: on a 1-GPU box a 4-task GPU topology would otherwise hand out
    # cuda:1..3 (nonexistent devices).
    n_gpus = constants.NODE_GPU_COUNT
    try:
        import torch
        if torch.cuda.is_available():
            n_gpus = max(1, torch.cuda.device_count())
    except Exception:  # noqa: BLE001 - torch-free client stays usable
        pass
    assignment: Dict[str, List[int]] = {}
    next_gpu = 0
    for t in tasks:
        spec = task_specs.get(t.type)
        key = f"{t.type}:{t.id}"
        if (spec is not None and spec.label == NodeLabel.GPU
                and t.type in ("chief", "worker", "ps")):
            gpus = [(next_gpu + i) % n_gpus for i in range(t.nb_proc)]
            next_gpu = (next_gpu + t.nb_proc) % n_gpus
            assignment[key] = gpus
        else:
            assignment[key] = []
    return assignment


def _task_env(task: ContainerTask,
              kv_addr: str,
              app_id: str,
              app_dir: str,
              gpus: List[int],
              n_try: int,
              task_specs: TaskSpecs,
              log_path: str,
              extra_env: Optional[Dict[str, str]] = None
              ) -> Dict[str, str]:
    """Build the per-container environment contract
    (reference ``client.py:108-133`` + TB_* plumbing ``client.py:213-219``)."""
    child = dict(os.environ)
    # Ship the framework itself into the task env (the reference zips
    # tf_yarn into each container, client.py:108-133): here it is enough to
    # put the package's parent directory on the child's PYTHONPATH.
    pkg_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    existing_pp = child.get("PYTHONPATH", "")
    if pkg_root not in existing_pp.split(os.pathsep):
        child["PYTHONPATH"] = (pkg_root + os.pathsep + existing_pp
                               if existing_pp else pkg_root)
    child[constants.ENV_CONTAINER_ID] = f"{task.type}_{task.id}"
    child[constants.ENV_KV_ADDR] = kv_addr
    child[constants.ENV_APP_ID] = app_id
    child[constants.ENV_APP_DIR] = app_dir
    child[constants.ENV_N_TRY] = str(n_try)
    child["MIYARN_GPU_IDS"] = ",".join(str(g) for g in gpus)
    child["MIYARN_LOG_FILE"] = log_path
    # vcores bound the container's CPU parallelism (the reference's YARN
    # vcore semantics, topologies.py:8-9) — and prevent N tasks x
    # all-cores OMP oversubscription on small hosts.
    spec = task_specs.get(task.type)
    if spec is not None:
        child.setdefault("OMP_NUM_THREADS", str(max(1, spec.vcores)))
    # RCCL over xGMI: dmabuf IPC only on this driver.
    child.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    if mlflow.use_mlflow:
        child["MLFLOW_TRACKING_URI"] = mlflow.get_tracking_uri() or ""
        child["MLFLOW_RUN_ID"] = mlflow.active_run_id() or ""
        child["GIT_PYTHON_REFRESH"] = "quiet"
    spec = task_specs.get(task.type)
    if task.type == "tensorboard" and spec is not None:
        child["TB_TERMINATION_TIMEOUT_SECONDS"] = str(
            spec.tb_termination_timeout_seconds)
        if spec.tb_model_dir:
            child["TB_MODEL_DIR"] = spec.tb_model_dir
        if spec.tb_extra_args:
            child["TB_EXTRA_ARGS"] = spec.tb_extra_args
    # user-supplied env (reference run_on_yarn(env={...}), client.py:306)
    if extra_env:
        child.update({k: str(v) for k, v in extra_env.items()})
    return child


def _spawn_tasks(tasks: List[ContainerTask],
                 task_specs: TaskSpecs,
                 server: KVServer,
                 app_id: str,
                 app_dir: str,
                 n_try: int,
                 custom_task_module: Optional[str],
                 pre_script_hook: Optional[str],
                 extra_env: Optional[Dict[str, str]] = None
                 ) -> List[TaskProcess]:
    gpu_map = _allocate_gpus(tasks, task_specs)
    log_dir = os.path.join(app_dir, "logs")
    os.makedirs(log_dir, exist_ok=True)
    processes = []
    for task in tasks:
        module = env_mod.gen_task_cmd(task.type, custom_task_module)
        log_path = os.path.join(log_dir, f"{task.type}_{task.id}.log")
        child_env = _task_env(task, server.address, app_id, app_dir,
                              gpu_map[f"{task.type}:{task.id}"], n_try,
                              task_specs, log_path, extra_env)
        cmd = [sys.executable, "-m", module]
        if pre_script_hook:
            cmd = ["bash", "-c",
                   f"{pre_script_hook} && exec {sys.executable} -m {module}"]
        logf = open(log_path, "wb")
        proc = subprocess.Popen(cmd, env=child_env, stdout=logf,
                                stderr=subprocess.STDOUT,
                                start_new_session=True)
        logf.close()
        processes.append(TaskProcess(task, proc, log_path))
        logger.info("spawned %s:%s pid=%d module=%s gpus=%s",
                    task.type, task.id, proc.pid, module,
                    child_env["MIYARN_GPU_IDS"])
    return processes


def _aggregate_events(client: KVClient,
                      events: Dict[str, Dict[str, str]]) -> None:
    """Event-listener thread body (reference ``client.py:633-657``):
    consume the KV watch stream and bucket events per task."""
    for key, value in client.events(""):
        if key.startswith("c10d/"):
            continue  # binary rendezvous traffic, not lifecycle events
        if "/" in key:
            task, stage = key.split("/", 1)
            events.setdefault(task, {})[stage] = value.decode(
                errors="replace")


def _setup_cluster(task_specs: TaskSpecs,
                   n_try: int,
                   custom_task_module: Optional[str],
                   pre_script_hook: Optional[str],
                   base_dir: Optional[str] = None,
                   extra_env: Optional[Dict[str, str]] = None
                   ) -> LocalCluster:
    """The ``_setup_skein_cluster`` equivalent (reference ``client.py:179``)."""
    app_id = f"miyarn_{uuid.uuid4().hex[:12]}"
    app_dir = os.path.join(base_dir or os.environ.get(
        "MIYARN_APP_BASE_DIR", "/tmp/miyarn"), app_id)
    os.makedirs(app_dir, exist_ok=True)
    server = KVServer()
    client = KVClient(server.address)
    tasks = iter_tasks(task_specs)
    _setup_cluster_spec(tasks, client)
    events: Dict[str, Dict[str, str]] = {}
    listener = threading.Thread(
        target=_aggregate_events, args=(KVClient(server.address), events),
        name="event-listener", daemon=True)
    listener.start()
    processes = _spawn_tasks(tasks, task_specs, server, app_id, app_dir,
                             n_try, custom_task_module, pre_script_hook,
                             extra_env)
    return LocalCluster(server, client, app_id, app_dir, tasks, processes,
                        listener, events)


def _shutdown_cluster(cluster: LocalCluster, sig=signal.SIGTERM) -> None:
    """Reference ``_shutdown_on_exception`` (``client.py:508-524``)."""
    for tp in cluster.processes:
        if tp.process.poll() is None:
            try:
                os.killpg(os.getpgid(tp.process.pid), sig)
            except (ProcessLookupError, PermissionError):
                pass
    deadline = time.time() + 10
    for tp in cluster.processes:
        try:
            tp.process.wait(timeout=max(0.1, deadline - time.time()))
        except subprocess.TimeoutExpired:
            try:
                os.killpg(os.getpgid(tp.process.pid), signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                pass
    cluster.server.stop()


def _execute_and_await_termination(
        cluster: LocalCluster,
        serialized_fn: bytes,
        eval_monitor_log_thresholds: Optional[Dict] = None,
        n_try: int = 0,
        poll_period: float = POLL_PERIOD_SECS) -> Tuple[Optional[Metrics],
                                                        ContainerLogStatus]:
    """Ship the experiment and poll children until done
    (reference ``client.py:527-599``)."""
    cluster.client.put(constants.KV_EXPERIMENT_FN, serialized_fn)

    evaluators = [f"{t.type}:{t.id}" for t in cluster.tasks
                  if t.type == "evaluator"]
    eval_logger = EvaluatorMetricsLogger(
        evaluators, cluster.client,
        eval_monitor_log_thresholds, n_try=n_try)
    one_shot = OneShotMetricsLogger(
        cluster.client,
        [(f"{t.type}:{t.id}/url", "tensorboard URL")
         for t in cluster.tasks if t.type == "tensorboard"],
        n_try=n_try)

    statuses: Dict[str, str] = {}
    while True:
        running = False
        failed = False
        for tp in cluster.processes:
            rc = tp.process.poll()
            if rc is None:
                running = True
            else:
                statuses.setdefault(
                    tp.key, "SUCCEEDED" if rc == 0 else "FAILED")
                if rc != 0:
                    failed = True
        eval_logger.log()
        one_shot.log()
        if failed:
            break
        if not running:
            break
        # Side tasks may outlive training by design: the evaluator keeps
        # scanning checkpoints (20-min idle cap), tensorboard lingers its
        # termination timeout, and ps exits once every worker says
        # goodbye + the stop-barrier releases it (reference
        # _independent_workers_task.py:38-43, tf_task_common.py:102-107).
        # The poll loop therefore waits for ALL processes.
        time.sleep(poll_period)

    if failed:
        # give surviving tasks a moment to publish their stop events
        time.sleep(1.0)
    _shutdown_cluster(cluster)
    for tp in cluster.processes:
        rc = tp.process.poll()
        statuses.setdefault(
            tp.key,
            "SUCCEEDED" if rc == 0 else
            ("KILLED" if rc is not None and rc < 0 else "FAILED"))

    log_paths = {tp.key: tp.log_path for tp in cluster.processes}
    container_status = ContainerLogStatus(log_paths, statuses)
    metrics = _handle_events(cluster.events, cluster.tasks)
    if any(s == "FAILED" for k, s in statuses.items()):
        _print_remote_tracebacks(cluster.events)
        raise RunFailed(
            "tasks failed: "
            + ", ".join(k for k, s in statuses.items() if s == "FAILED"))
    return metrics, container_status


def _print_remote_tracebacks(events: Dict[str, Dict[str, str]]) -> None:
    """Reference ``client.py:723-725``."""
    for task, stages in sorted(events.items()):
        exc = stages.get("stop")
        if exc:
            logger.error("%s failed with:\n%s", task, exc)


def _handle_events(events: Dict[str, Dict[str, str]],
                   tasks: List[ContainerTask]) -> Metrics:
    """Compute wall-time metrics from lifecycle events
    (reference ``client.py:692-739``: training time =
    max(train_eval_stop) - min(train_eval_start) across chief+workers)."""
    def _f(task: str, stage: str) -> Optional[float]:
        v = events.get(task, {}).get(stage)
        try:
            return float(v) if v else None
        except ValueError:
            return None

    train_tasks = [f"{t.type}:{t.id}" for t in tasks
                   if t.type in ("chief", "worker")]
    eval_tasks = [f"{t.type}:{t.id}" for t in tasks
                  if t.type == "evaluator"]

    def _span(task_list: List[str]) -> Optional[float]:
        starts = [x for x in
                  (_f(t, event.TRAIN_EVAL_START_TIME) for t in task_list)
                  if x is not None]
        stops = [x for x in
                 (_f(t, event.TRAIN_EVAL_STOP_TIME) for t in task_list)
                 if x is not None]
        if not starts or not stops:
            return None
        return max(stops) - min(starts)

    container_duration = {}
    train_eval_time = {}
    for t in tasks:
        key = f"{t.type}:{t.id}"
        start = _f(key, event.CONTAINER_START_TIME)
        stop = _f(key, event.CONTAINER_STOP_TIME)
        container_duration[key] = (
            stop - start if start is not None and stop is not None else None)
        s0 = _f(key, event.TRAIN_EVAL_START_TIME)
        s1 = _f(key, event.TRAIN_EVAL_STOP_TIME)
        train_eval_time[key] = (
            s1 - s0 if s0 is not None and s1 is not None else None)

    return Metrics(_span(train_tasks), _span(eval_tasks),
                   container_duration, train_eval_time)


def run_on_yarn(experiment_fn,
                task_specs: Optional[TaskSpecs] = None,
                *,
                custom_task_module: Optional[str] = None,
                nb_retries: int = 0,
                pre_script_hook: Optional[str] = None,
                eval_monitor_log_thresholds: Optional[Dict] = None,
                name: str = "RunOnMI355X",
                queue: str = "default",
                ps_strategy: bool = False,
                base_dir: Optional[str] = None,
                env: Optional[Dict[str, str]] = None,
                **_ignored) -> Optional[Metrics]:
    """Launch a distributed experiment on the local MI355X node.

    API parity with the reference's ``run_on_yarn`` (``client.py:299-470``):
    same task-spec topology model, cloudpickled ``experiment_fn``, whole-run
    retry loop, event aggregation to :class:`Metrics`, ``RunFailed`` on task
    failure.  ``env`` adds user variables to every task's environment (reference
    ``client.py:306``); ``name``/``queue`` are accepted for compatibility
    and unused (there is no YARN queue on one node).
    """
    if task_specs is None:
        task_specs = {"chief": TaskSpec()}
    if ps_strategy or "ps" in task_specs:
        _check_ps_topology(task_specs)
    else:
        # custom task modules define their own topology needs (e.g. the
        # reference README's evaluation-only run, README.md:371-380)
        _check_general_topology(task_specs,
                                require_chief=custom_task_module is None)

    with catchtime("serializing experiment_fn"):
        serialized_fn = cloudpickle.dumps(experiment_fn)

    n_try = 0
    while True:
        try:
            with catchtime(f"setting up cluster (try {n_try})"):
                cluster = _setup_cluster(
                    task_specs, n_try, custom_task_module, pre_script_hook,
                    base_dir, env)
            try:
                metrics, container_status = _execute_and_await_termination(
                    cluster, serialized_fn,
                    eval_monitor_log_thresholds, n_try)
            except BaseException:
                _shutdown_cluster(cluster)
                raise
            if metrics is not None:
                metrics.log_mlflow(n_try)
            _log_container_tails(container_status)
            return metrics
        except RunFailed:
            if n_try < nb_retries:
                logger.warning("run failed, retrying (%d/%d)",
                               n_try + 1, nb_retries)
                n_try += 1
                continue
            raise
        except KeyboardInterrupt:
            raise


def _log_container_tails(container_status: ContainerLogStatus,
                         max_bytes: int = 64 * 1024) -> None:
    """Attach per-container log tails to mlflow
    (reference ``client.py:605-617,748-765``)."""
    for task, (path, status) in container_status.by_container_id().items():
        if not path or not os.path.exists(path):
            continue
        try:
            with open(path, "rb") as f:
                f.seek(0, os.SEEK_END)
                size = f.tell()
                f.seek(max(0, size - max_bytes))
                tail = f.read().decode(errors="replace")
            mlflow.save_text_to_mlflow(
                tail, f"{mlflow.format_key(task)}_{status}.log")
        except OSError:
            pass
