"""Shared constants.

Mirrors the semantics of the reference's constant surface
(elasticai_api/common/constants.py:15-43) without copying it: gRPC message
caps, environment-variable channel between master and workers/PS.
"""

# gRPC channel limits: the PS data plane moves whole dense models and
# embedding batches in single messages (reference caps these at 256 MB).
GRPC_MAX_MESSAGE_BYTES = 256 * 1024 * 1024

GRPC_CHANNEL_OPTIONS = [
    ("grpc.max_send_message_length", GRPC_MAX_MESSAGE_BYTES),
    ("grpc.max_receive_message_length", GRPC_MAX_MESSAGE_BYTES),
]


class WorkerEnv:
    """Env-var channel master -> worker/PS (reference: constants.py:38-43)."""

    MASTER_ADDR = "EDL_MASTER_ADDR"
    WORKER_ID = "EDL_WORKER_ID"
    WORKER_NUM = "EDL_WORKER_NUM"
    POD_IP = "EDL_POD_IP"
    PS_ADDRS = "EDL_PS_ADDRS"


class DistributionStrategy:
    LOCAL = "Local"
    PARAMETER_SERVER = "ParameterServerStrategy"
    ALLREDUCE = "AllreduceStrategy"


class JobType:
    TRAINING_ONLY = "training_only"
    EVALUATION_ONLY = "evaluation_only"
    PREDICTION_ONLY = "prediction_only"
    TRAINING_WITH_EVALUATION = "training_with_evaluation"


class TaskExecCounterKey:
    FAIL_COUNT = "fail_count"


class PodStatus:
    INITIAL = "Initial"
    PENDING = "Pending"
    RUNNING = "Running"
    SUCCEEDED = "Succeeded"
    FAILED = "Failed"
    DELETED = "Deleted"
    UNKNOWN = "Unknown"


# Maximum times a task is re-dispatched after worker failure
# (reference: task_manager.py:31).
MAX_TASK_RETRIES = 3

# Maximum times one minibatch is retried inside a worker
# (reference: worker/worker.py:39).
MAX_MINIBATCH_RETRY_NUM = 64

# Allreduce communicator re-init retries on collective failure
# (reference: allreduce_trainer.py:66-91).
MAX_ALLREDUCE_RETRY_NUM = 5
