"""tf_yarn_amd — MI355X-native distributed-training launcher.

A brand-new single-node framework with the capabilities of criteo/tf-yarn
(reference mounted at /root/reference): the same ``run_on_yarn`` /
``experiment_fn`` API and role topologies, with YARN containers replaced by
local processes pinned to 8 MI355X GPUs, the skein KV store replaced by an
own control-plane store, and the gradient data planes (DDP allreduce,
Horovod-style optimizer, parameter servers) re-implemented on RCCL over xGMI
with hand-written CDNA4 HIP kernels.

Framework-agnostic core only here (the reference's ``tf_yarn/__init__.py``
likewise never imports TF or torch); flavors live in
``tf_yarn_amd.pytorch``, ``tf_yarn_amd.estimator`` and
``tf_yarn_amd.distributed``.
"""

from tf_yarn_amd.client import (RunFailed, get_safe_experiment_fn,
                                run_on_yarn)
from tf_yarn_amd.metrics import Metrics
from tf_yarn_amd.topologies import (ContainerKey, ContainerTask, NodeLabel,
                                    TaskSpec, ps_strategy_topology,
                                    single_server_topology)

__version__ = "0.1.0"

__all__ = [
    "run_on_yarn", "RunFailed", "get_safe_experiment_fn", "Metrics",
    "TaskSpec", "NodeLabel", "ContainerKey", "ContainerTask",
    "single_server_topology", "ps_strategy_topology",
]
