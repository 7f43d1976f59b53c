"""Wire-contract constants for the AITrainingJob API.

Byte-compatible with the reference operator's constants
(reference: pkg/apis/aitrainingjob/v1/constants.go:1-77 and register.go:27-33).
Every label key, env-var name, container/port prefix, error status and
phase->reason mapping the reference emits is reproduced here so that workloads
written against the reference keep working unmodified.

MI355X extension: additional rendezvous env vars (MASTER_ADDR/..., RANK, ...)
are defined at the bottom; they are *additive* — the reference contract is a
strict subset of what this operator injects.
"""

# --- CRD identity (reference: pkg/apis/aitrainingjob/v1/register.go:27-33) ---
CRD_GROUP = "elasticdeeplearning.ai"
CRD_VERSION = "v1"
CRD_KIND = "AITrainingJob"
CRD_KIND_LIST = "AITrainingJobList"
CRD_PLURAL = "aitrainingjobs"
CRD_SINGULAR = "aitrainingjob"
CRD_SHORT_NAME = "aitj"
CRD_NAME = f"{CRD_PLURAL}.{CRD_GROUP}"
API_VERSION = f"{CRD_GROUP}/{CRD_VERSION}"

CONTROLLER_NAME = "TrainingJobOperator"

# --- Pod/Service labels (reference: constants.go:3-11) ---
LABEL_REPLICA_NAME = "TrainingJobReplicaName"
LABEL_REPLICA_INDEX = "TrainingJobReplicaIndex"
LABEL_JOB_NAME = "TrainingJobName"
LABEL_FRAMEWORK = "FrameworkType"
LABEL_GROUP_NAME = "GroupName"
LABEL_PRIORITY = "priority"
# Additional labels the reference stamps on pods (pkg/controller/pod.go:496-505)
LABEL_JOB_NAME_SHORT = "JobName"
LABEL_POD_ROLE = "PodRole"
LABEL_RESTART_COUNT = "RestartCount"
POD_ROLE_VALUE = "container"

# --- Injected env vars (reference: constants.go:13-21) ---
ENV_REPLICA_NAME = "TRAININGJOB_REPLICA_NAME"
ENV_REPLICA_INDEX = "TRAININGJOB_REPLICA_INDEX"
ENV_REPLICA_RESTART_COUNT = "TRAININGJOB_REPLICA_RESTARTCOUNT"
ENV_JOB_NAME = "TRAININGJOB_NAME"
ENV_JOB_NAMESPACE = "TRAININGJOB_NAMESPACE"
ENV_SERVICE = "TRAININGJOB_SERVICE"
ENV_PORTS = "TRAININGJOB_PORTS"

# --- MI355X-native rendezvous extension (additive) ---
# Injected so stock torchrun / torch.distributed workers bootstrap RCCL
# (backend "nccl" on ROCm) with zero launcher glue.
ENV_MASTER_ADDR = "MASTER_ADDR"
ENV_MASTER_PORT = "MASTER_PORT"
ENV_WORLD_SIZE = "WORLD_SIZE"
ENV_RANK = "RANK"
ENV_LOCAL_RANK = "LOCAL_RANK"
ENV_REND_EPOCH = "TRAININGJOB_RENDEZVOUS_EPOCH"  # bumped on every elastic resize
ENV_LOCAL_WORLD_SIZE = "LOCAL_WORLD_SIZE"
ENV_NPROC_PER_NODE = "NPROC_PER_NODE"       # torchrun convention
ENV_NODE_RANK = "NODE_RANK"
GPU_RESOURCE = "amd.com/gpu"                # ROCm k8s device plugin
ENV_MIN_REPLICAS = "TRAININGJOB_MIN_REPLICAS"
ENV_MAX_REPLICAS = "TRAININGJOB_MAX_REPLICAS"
DEFAULT_MASTER_PORT = 23456

# ROCm GPU resource name served by the ROCm k8s device plugin.
AMD_GPU_RESOURCE = "amd.com/gpu"
# HBM3E capacity per MI355X GPU, used by HBM-aware replica sizing.
MI355X_HBM_BYTES = 288 * (1 << 30)

# --- Container / port participation (reference: constants.go:41-44) ---
CONTAINER_PREFIX = "aitj-"
PORT_PREFIX = "aitj-"

# --- Condition/event reasons (reference: constants.go:25-39) ---
POD_TEMPLATE_RESTART_POLICY_REASON = "SettedPodTemplateRestartPolicy"
EXITED_WITH_CODE_REASON = "ExitedWithCode"

# Waiting-state container reasons treated as creation errors
# (reference: constants.go:47-56).
ERROR_CONTAINER_STATUS = [
    "CreateContainerConfigError",
    "CreateContainerError",
    "ImagePullBackOff",
    "ImageInspectError",
    "ErrImagePull",
    "ErrImageNeverPull",
    "RegistryUnavailable",
    "InvalidImageName",
]
