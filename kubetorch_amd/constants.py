"""Framework-wide constants (reference parity: python_client/kubetorch/constants.py)."""

# ports
SERVER_PORT = 32300          # worker pod HTTP server
CONTROLLER_PORT = 8081       # controller API + pod WebSocket hub
NGINX_PORT = 8080            # public front proxy
DATA_STORE_PORT = 8873       # data-store file/metadata server
LOG_STORE_PORT = 3100        # log store (push + WS tail)
METRICS_PORT = 9090          # metrics store
GPU_DATA_SERVER_TCP_PORT = 29400   # pod-data-server server<->server
RCCL_PORT_RANGE = (29500, 29600)   # RCCL rendezvous ports for data-plane PGs
DEFAULT_MASTER_PORT = 12355  # SPMD torch.distributed rendezvous

# k8s labels / annotations (amd.com/gpu scheduling, not nvidia)
LABEL_PREFIX = "kubetorch.amd.com"
SERVICE_LABEL = f"{LABEL_PREFIX}/service"
VERSION_LABEL = f"{LABEL_PREFIX}/version"
MODULE_LABEL = f"{LABEL_PREFIX}/module"
USERNAME_LABEL = f"{LABEL_PREFIX}/username"
INACTIVITY_TTL_ANNOTATION = f"{LABEL_PREFIX}/inactivity-ttl"
KUEUE_QUEUE_LABEL = "kueue.x-k8s.io/queue-name"
GPU_RESOURCE = "amd.com/gpu"
GPU_PRODUCT_LABEL = "amd.com/gpu.product-name"

# timeouts (seconds)
LAUNCH_TIMEOUT = 900
HTTP_TIMEOUT = 120
QUORUM_TIMEOUT = 300
RCCL_TRANSFER_TIMEOUT = 60
DNS_MONITOR_INTERVAL = 3.0
RELOAD_ACK_TIMEOUT = 120

# env var names (worker pod metadata contract)
ENV_MODULE_NAME = "KT_MODULE_NAME"
ENV_CALLABLE_NAME = "KT_CLS_OR_FN_NAME"
ENV_FILE_PATH = "KT_FILE_PATH"
ENV_PROJECT_ROOT = "KT_PROJECT_ROOT"
ENV_INIT_ARGS = "KT_INIT_ARGS"
ENV_DISTRIBUTED_CONFIG = "KT_DISTRIBUTED_CONFIG"
ENV_SERVICE_NAME = "KT_SERVICE_NAME"
ENV_SERVICE_DNS = "KT_SERVICE_DNS"
ENV_LAUNCH_ID = "KT_LAUNCH_ID"
ENV_LOCAL_IPS = "KT_LOCAL_IPS"   # fake-cluster mode for tests/local driver
ENV_MODULE_TYPE = "KT_MODULE_TYPE"

TERMINATION_REASONS = ("Evicted", "OOMKilled", "DeadlineExceeded", "Preempted")

# RCCL / xGMI tuning defaults for intra-node 8x MI355X (set by launchers)
RCCL_ENV_DEFAULTS = {
    "HSA_ENABLE_IPC_MODE_LEGACY": "0",   # host driver supports dmabuf IPC only
    "NCCL_IB_DISABLE": "1",              # intra-node xGMI, no IB
    "TORCH_NCCL_ASYNC_ERROR_HANDLING": "1",
}
