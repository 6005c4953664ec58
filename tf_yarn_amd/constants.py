"""Key-value store key constants.

Parity with the reference's ``tf_yarn/constants.py:1-3`` (``cluster_instances``,
``experiment_fn``, session-config keys) plus the env-var vocabulary that replaces
the YARN/skein container environment.
"""

KV_CLUSTER_INSTANCES = "cluster_instances"
KV_EXPERIMENT_FN = "experiment_fn"
KV_TF_SESSION_CONFIG = "tf_session_config"

# Environment variables set by the spawner in every task process
# (the SKEIN_CONTAINER_ID / MASTER_ADDR contract of the reference,
# _task_commons.py:15-16, client.py:108-133).
ENV_CONTAINER_ID = "MIYARN_CONTAINER_ID"
ENV_KV_ADDR = "MIYARN_KV_ADDR"
ENV_APP_ID = "MIYARN_APP_ID"
ENV_APP_DIR = "MIYARN_APP_DIR"
ENV_N_TRY = "TF_YARN_N_TRY"
ENV_MASTER_ADDR = "MASTER_ADDR"
ENV_MASTER_PORT = "MASTER_PORT"

# Number of GPUs on one MI355X node.
NODE_GPU_COUNT = 8
