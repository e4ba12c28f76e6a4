"""Public API surface of the kubeshare-amd stack: pod labels, annotations,
environment variables, on-disk paths and operational defaults.

The label vocabulary is the reference's public workload API and is kept
verbatim (reference: pkg/scheduler/constants.go:3-28, README.md:32-105).
Everything NVIDIA-specific is replaced by the ROCm-native equivalent
(env injection: reference pkg/scheduler/pod.go:435-476).
"""

# ---------------------------------------------------------------- labels (L5)
DOMAIN = "sharedgpu/"

POD_GROUP_NAME = DOMAIN + "group_name"
POD_GROUP_HEADCOUNT = DOMAIN + "group_headcount"
POD_GROUP_THRESHOLD = DOMAIN + "group_threshold"
# The aggregator vocabulary for gang size; the reference accepts it in
# aggregator/pod.go:22 while the scheduler derives it from
# headcount*threshold (pod_group.go:86-117). We accept both (SURVEY.md
# Appendix A) and document min_available as the canonical exported form.
POD_MIN_AVAILABLE = DOMAIN + "min_available"

POD_PRIORITY = DOMAIN + "priority"
POD_GPU_LIMIT = DOMAIN + "gpu_limit"
POD_GPU_REQUEST = DOMAIN + "gpu_request"
POD_GPU_MEMORY = DOMAIN + "gpu_mem"
POD_GPU_MODEL = DOMAIN + "gpu_model"
# optional lease-length bound in ms: caps how long THIS pod's leases
# can block co-located pods (a request arriving mid-lease waits for the
# holder's drain). Set a small value on throughput pods sharing a GPU
# with latency-critical serving — the serving pod's own leases are
# already right-sized by its busy-EWMA hint. Clamped server-side to
# [min_quota, base_quota]; BASELINE.md quantifies the tradeoff.
POD_LEASE_MS = DOMAIN + "lease_ms"

# annotations written by Reserve (reference pod.go:402-476)
POD_GPU_UUID = DOMAIN + "gpu_uuid"
POD_CELL_ID = DOMAIN + "cell_id"
POD_MANAGER_PORT = DOMAIN + "gpu_manager_port"
# node-local device index(es) for ROCR_VISIBLE_DEVICES (consumed by the
# mutating webhook, kubeshare_amd/webhook.py)
POD_GPU_INDEX = "kubeshare.amd/gpu_index"

SCHEDULER_NAME = "kubeshare-scheduler"

# ------------------------------------------------------------- env vars (L1)
# Injected into shared-GPU containers (ROCm-native replacements for the
# reference's NVIDIA_VISIBLE_DEVICES / LD_PRELOAD block, pod.go:445-457).
ENV_ROCR_VISIBLE_DEVICES = "ROCR_VISIBLE_DEVICES"
ENV_HIP_VISIBLE_DEVICES = "HIP_VISIBLE_DEVICES"
ENV_LD_PRELOAD = "LD_PRELOAD"
ENV_POD_MANAGER_IP = "POD_MANAGER_IP"
ENV_POD_MANAGER_PORT = "POD_MANAGER_PORT"
ENV_POD_MANAGER_UDS = "POD_MANAGER_UDS"
# marker the webhook keys idempotency on (a user-set ROCR_VISIBLE_DEVICES
# must NOT suppress injection — that would silently skip the hook)
ENV_INJECTED = "KUBESHARE_INJECTED"
ENV_POD_NAME = "POD_NAME"
ENV_SCHEDULER_IP = "SCHEDULER_IP"
ENV_SCHEDULER_PORT = "SCHEDULER_PORT"
# memory cap in bytes enforced by libhiphook (default request * full memory)
ENV_GPU_MEM = "KUBESHARE_GPU_MEM"
# abort instead of silently running un-gated when the hook failed to attach
ENV_REQUIRE_HOOK = "KUBESHARE_REQUIRE_HOOK"

# ------------------------------------------------------------ hostPath (L2)
KUBESHARE_ROOT = "/kubeshare"
LIBRARY_PATH = KUBESHARE_ROOT + "/library"
HOOK_SO_NAME = "libhiphook.so"
HOOK_SO_PATH = LIBRARY_PATH + "/" + HOOK_SO_NAME
SCHEDULER_IP_FILE = LIBRARY_PATH + "/schedulerIP.txt"
# per-pod pod-mgr unix sockets live here (hostPath mounted RW into the
# container; a UDS connect() needs write access to the socket inode, so
# this cannot share the read-only /kubeshare/library mount). Default
# transport hook<->pod-mgr: UDS keyed by the pod's manager port — no
# hostNetwork, unreachable from other pods that don't mount it.
SOCK_DIR = KUBESHARE_ROOT + "/sock"


def pod_manager_uds(port: int, root: str = SOCK_DIR) -> str:
    """Socket path for one pod's manager, keyed by its allocated port
    (ports are already unique per node via the 512-wide pool)."""
    return f"{root}/pm-{port}.sock"
LOG_PATH = KUBESHARE_ROOT + "/log"
SCHEDULER_CONFIG_ROOT = KUBESHARE_ROOT + "/scheduler"
GPU_CONFIG_DIR = SCHEDULER_CONFIG_ROOT + "/config/"
POD_MANAGER_PORT_DIR = SCHEDULER_CONFIG_ROOT + "/podmanagerport/"
CLUSTER_TOPOLOGY_FILE = SCHEDULER_CONFIG_ROOT + "/kubeshare-config.yaml"

# ------------------------------------------------------------- defaults (L1)
# Token-scheduler knobs; values are the reference's operational defaults
# (docker/kubeshare-gemini-scheduler/launcher.py:77-80).
BASE_QUOTA_MS = 300.0
MIN_QUOTA_MS = 20.0
WINDOW_MS = 10000.0

# gpu-schd listens on BASE_SCHED_PORT + gpu_index
# (launcher-multigpus.sh:21,41); pod managers draw from a 512-wide pool
# (pkg/scheduler/node.go:13-15, scheduler.go:348-360).
BASE_SCHED_PORT = 49901
POD_MANAGER_PORT_START = 50050
POD_MANAGER_PORT_POOL = 512

# gang semantics (reference scheduler.go:44-47)
PERMIT_WAITING_TIME_SEC = 2  # * group headcount
POD_GROUP_GC_INTERVAL_SEC = 30
POD_GROUP_EXPIRATION_SEC = 600

# --------------------------------------------------------------- MI355X (HW)
MI355X_MODEL = "AMD Instinct MI355X"
MI355X_GFX = "gfx950"
MI355X_HBM_BYTES = 288 * 1024**3  # 288 GiB HBM3E
MI355X_GPUS_PER_NODE = 8
MI355X_XGMI_LINKS_PER_GPU = 7  # point-to-point, ~153 GB/s each
MI355X_XGMI_LINK_GBPS = 153.0

# ------------------------------------------------------------- telemetry (L3)
METRIC_GPU_CAPACITY = "gpu_capacity"       # reference pkg/collector/collector.go:30-35
METRIC_GPU_REQUIREMENT = "gpu_requirement"  # reference pkg/aggregator/aggregator.go:22-39
COLLECTOR_PORT = 9004
AGGREGATOR_PORT = 9005
