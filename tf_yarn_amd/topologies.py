"""Task-role resource model and topology validation.

Parity with the reference's ``tf_yarn/topologies.py``: the same
``TaskSpec`` field names (``memory``/``vcores`` kept for API compatibility,
re-interpreted for one MI355X node), the five task roles, the
``ContainerKey``/``ContainerTask`` identity model and the two topology
factories (``topologies.py:130,144``).  Differences from the reference are
deliberate: containers become local processes, ``NodeLabel.GPU`` pins a task
to MI355X GPUs, and limits reflect one node (8 GPUs) instead of YARN caps.
"""

from enum import Enum
from typing import Dict, NamedTuple, Optional

from tf_yarn_amd import constants

# Single-node ceilings (the reference caps at 48 GB / 48 vcores per YARN
# container, topologies.py:8-9; here the unit is one MI355X node).
MAX_MEMORY_MB = 2 * 1024 * 1024  # 2 TB host RAM guard
MAX_VCORES = 256

ALL_TASK_TYPES = {"chief", "worker", "ps", "evaluator", "tensorboard"}
TRAINING_TASK_TYPES = {"chief", "worker"}
GPU_ELIGIBLE_TASK_TYPES = {"chief", "worker", "ps"}


class NodeLabel(Enum):
    """Hardware constraint for a task role (reference ``topologies.py:16``)."""
    CPU = ""
    GPU = "gpu"


class ContainerKey(NamedTuple):
    """Identity of one task instance (reference ``topologies.py:26-40``)."""
    type: str
    id: int

    def to_kv_str(self) -> str:
        return f"{self.type}:{self.id}"

    @classmethod
    def from_kv_str(cls, s: str) -> "ContainerKey":
        t, i = s.split(":")
        return cls(t, int(i))


class ContainerTask(NamedTuple):
    """One task instance plus its local process count
    (reference ``topologies.py:42-52``)."""
    type: str
    id: int
    nb_proc: int

    def to_container_key(self) -> ContainerKey:
        return ContainerKey(self.type, self.id)


def _parse_memory(value) -> int:
    """Memory in MiB from an int (MiB) or a skein-style string like
    ``"2 GiB"`` / ``"512 MiB"`` / ``"1 GB"`` (the reference's TaskSpec
    accepts ``Union[int, str]`` via ``skein.model.Resources``,
    ``topologies.py:64-72``)."""
    if isinstance(value, (int, float)):
        return int(value)
    text = str(value).strip().lower()
    units = {"kib": 1 / 1024, "kb": 1 / 1024, "k": 1 / 1024,
             "mib": 1, "mb": 1, "m": 1,
             "gib": 1024, "gb": 1024, "g": 1024,
             "tib": 1024 * 1024, "tb": 1024 * 1024, "t": 1024 * 1024,
             "b": 1 / (1024 * 1024)}
    for suffix in sorted(units, key=len, reverse=True):
        if text.endswith(suffix):
            number = text[:-len(suffix)].strip()
            return int(float(number) * units[suffix])
    return int(float(text))  # bare number string: MiB


class TaskSpec:
    """Resources requested for all instances of one task role.

    Field names mirror the reference (``topologies.py:54-95``):
    ``memory`` (MiB, or a skein-style string like ``"2 GiB"``),
    ``vcores``, ``instances``, ``nb_proc_per_worker``,
    ``label``, ``tb_termination_timeout_seconds``, ``tb_model_dir``,
    ``tb_extra_args``.  ``nb_proc_per_worker`` is the number of training
    processes (one per GPU for GPU-labelled tasks) per instance.
    """

    def __init__(self,
                 memory=1024,
                 vcores: int = 1,
                 instances: int = 1,
                 nb_proc_per_worker: int = 1,
                 label: NodeLabel = NodeLabel.CPU,
                 tb_termination_timeout_seconds: int = -1,
                 tb_model_dir: Optional[str] = None,
                 tb_extra_args: Optional[str] = None):
        self.memory = _parse_memory(memory)
        self.vcores = vcores
        self.instances = instances
        self.nb_proc_per_worker = nb_proc_per_worker
        self.label = label
        self.tb_termination_timeout_seconds = tb_termination_timeout_seconds
        self.tb_model_dir = tb_model_dir
        self.tb_extra_args = tb_extra_args

    def __repr__(self) -> str:
        return (f"TaskSpec(memory={self.memory}, vcores={self.vcores}, "
                f"instances={self.instances}, "
                f"nb_proc_per_worker={self.nb_proc_per_worker}, "
                f"label={self.label})")


TaskSpecs = Dict[str, TaskSpec]


def _check_general_topology(task_specs: TaskSpecs,
                            require_chief: bool = True) -> None:
    """Reference ``topologies.py:97-116``.  ``require_chief=False`` for
    custom task modules: the reference only validates topologies built by
    its factory functions, so e.g. the README's evaluation-only flow
    (``task_specs={"evaluator": ...}, custom_task_module=...``,
    ``README.md:371-380``) must be accepted."""
    unknown = set(task_specs) - ALL_TASK_TYPES
    if unknown:
        raise ValueError(
            f"unknown task types {sorted(unknown)}; "
            f"supported: {sorted(ALL_TASK_TYPES)}")
    if require_chief and (
            "chief" not in task_specs
            or task_specs["chief"].instances != 1):
        raise ValueError("exactly one chief task is required")
    for task_type, spec in task_specs.items():
        if spec.instances < 0:
            raise ValueError(f"{task_type}: instances must be >= 0")
        if spec.memory > MAX_MEMORY_MB or spec.vcores > MAX_VCORES:
            raise ValueError(
                f"{task_type}: requested {spec.memory} MiB / {spec.vcores} "
                f"vcores exceeds node limits "
                f"({MAX_MEMORY_MB} MiB / {MAX_VCORES} vcores)")
        if spec.nb_proc_per_worker > spec.vcores:
            raise ValueError(
                f"{task_type}: nb_proc_per_worker "
                f"({spec.nb_proc_per_worker}) must be <= vcores "
                f"({spec.vcores})")
    n_gpu_procs = sum(
        spec.instances * spec.nb_proc_per_worker
        for t, spec in task_specs.items()
        if spec.label == NodeLabel.GPU and t in GPU_ELIGIBLE_TASK_TYPES)
    if n_gpu_procs > constants.NODE_GPU_COUNT:
        raise ValueError(
            f"{n_gpu_procs} GPU training processes requested but the node "
            f"has {constants.NODE_GPU_COUNT} GPUs")


def _check_ps_topology(task_specs: TaskSpecs) -> None:
    """Reference ``topologies.py:118-128``."""
    _check_general_topology(task_specs)
    if task_specs.get("evaluator", TaskSpec(instances=0)).instances > 1:
        raise ValueError("at most one evaluator is supported")
    if task_specs.get("tensorboard", TaskSpec(instances=0)).instances > 1:
        raise ValueError("at most one tensorboard is supported")
    if "ps" in task_specs and task_specs["ps"].instances < 1:
        raise ValueError("ps strategy requires at least one ps task")


def single_server_topology(memory: int = 4096,
                           vcores: int = 1) -> TaskSpecs:
    """One chief + one evaluator (reference ``topologies.py:130-142``)."""
    return {
        "chief": TaskSpec(memory=memory, vcores=vcores),
        "evaluator": TaskSpec(memory=memory, vcores=vcores),
    }


def ps_strategy_topology(nb_workers: int = 2,
                         nb_ps: int = 1,
                         memory: int = 4096,
                         vcores: int = 1) -> TaskSpecs:
    """Parameter-server topology (reference ``topologies.py:144-160``)."""
    return {
        "chief": TaskSpec(memory=memory, vcores=vcores),
        "worker": TaskSpec(memory=memory, vcores=vcores,
                           instances=nb_workers),
        "ps": TaskSpec(memory=memory, vcores=vcores, instances=nb_ps),
        "evaluator": TaskSpec(memory=memory, vcores=vcores),
    }
