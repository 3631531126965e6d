"""Constants: resource names, annotations, labels, envs, paths.

API-surface parity with reference pkg/util/consts.go, re-domained for
AMD (`amd.com` default, overridable — the reference supports --domain
the same way, consts.go:158-176).  Annotation suffixes and claim
formats are kept byte-compatible so operator tooling ports unchanged.
"""
from __future__ import annotations

COMPONENT_NAME = "vgpu-manager"
AMD_DOMAIN = "amd.com"

_domain = AMD_DOMAIN


def set_domain(domain: str) -> None:
    global _domain
    _domain = domain


def domain() -> str:
    return _domain


# ---- resource names ----
def vgpu_number_resource() -> str:
    return f"{_domain}/vgpu-number"


def vgpu_memory_resource() -> str:
    return f"{_domain}/vgpu-memory"


def vgpu_core_resource() -> str:
    return f"{_domain}/vgpu-cores"


def cpx_resource_prefix() -> str:
    """CPX compute partitions are the MIG analog on MI355X."""
    return f"{_domain}/cpx-"


# ---- node annotations ----
def node_heartbeat_ann() -> str:
    return f"{_domain}/node-device-heartbeat"


def node_register_ann() -> str:
    return f"{_domain}/node-device-register"


def node_topology_ann() -> str:
    return f"{_domain}/node-device-topology"


def node_config_ann() -> str:
    return f"{_domain}/node-config-info"


def node_gpu_domain_ann() -> str:
    return f"{_domain}/node-gpu-domain"


# ---- pod annotations / labels ----
def pre_alloc_ann() -> str:
    return f"{_domain}/pre-allocated"


def real_alloc_ann() -> str:
    return f"{_domain}/real-allocated"


def predicate_node_ann() -> str:
    return f"{_domain}/predicate-node"


def predicate_time_ann() -> str:
    return f"{_domain}/predicate-time"


def assigned_phase_label() -> str:
    return f"{_domain}/assigned-phase"


def node_scheduler_policy_ann() -> str:
    return f"{_domain}/node-scheduler-policy"


def device_scheduler_policy_ann() -> str:
    return f"{_domain}/device-scheduler-policy"


def memory_scheduler_policy_ann() -> str:
    return f"{_domain}/memory-scheduler-policy"


def topology_mode_ann() -> str:
    return f"{_domain}/device-topology-mode"


def cross_pod_topology_ann() -> str:
    return f"{_domain}/cross-pod-topology"


def include_gpu_uuid_ann() -> str:
    return f"{_domain}/include-gpu-uuid"


def exclude_gpu_uuid_ann() -> str:
    return f"{_domain}/exclude-gpu-uuid"


def include_gpu_type_ann() -> str:
    return f"{_domain}/include-gpu-type"


def exclude_gpu_type_ann() -> str:
    return f"{_domain}/exclude-gpu-type"


def compute_policy_ann() -> str:
    return f"{_domain}/vgpu-compute-policy"


def stuck_grace_period_ann() -> str:
    return f"{_domain}/stuck-grace-period"


IGNORE_WEBHOOK_ANN = "vgpu-manager.io/ignore-webhook"
DRA_ORIGINAL_RESOURCES_ANN = "vgpu-manager.io/original-resources"

# assigned-phase values
PHASE_ALLOCATING = "allocating"
PHASE_SUCCESS = "success"
PHASE_FAILED = "failed"

# scheduler policies
POLICY_BINPACK = "binpack"
POLICY_SPREAD = "spread"

# topology modes
TOPO_NONE = "none"
TOPO_NUMA = "numa"
TOPO_NUMA_STRICT = "numa-strict"
TOPO_LINK = "link"
TOPO_LINK_STRICT = "link-strict"

# compute policies
COMPUTE_FIXED = "fixed"
COMPUTE_BALANCE = "balance"
COMPUTE_NONE = "none"

# ---- container envs written by Allocate (shim-consumed) ----
ENV_MEM_LIMIT = "VGPU_MEM_LIMIT_{}"
ENV_CORE_LIMIT = "VGPU_CORE_LIMIT_{}"
ENV_CORE_SOFT_LIMIT = "VGPU_CORE_SOFT_LIMIT_{}"
ENV_MEM_OVERSOLD = "VGPU_MEM_OVERSOLD"
ENV_COMPUTE_POLICY = "VGPU_COMPUTE_POLICY"
ENV_POD_NAME = "VGPU_POD_NAME"
ENV_POD_NAMESPACE = "VGPU_POD_NAMESPACE"
ENV_POD_UID = "VGPU_POD_UID"
ENV_CONTAINER_NAME = "VGPU_CONTAINER_NAME"
ENV_DISABLE_CONTROL = "DISABLE_VGPU_CONTROL"
ENV_VISIBLE_DEVICES = "ROCR_VISIBLE_DEVICES"
ENV_MANAGER_VISIBLE_DEVICES = "MANAGER_VISIBLE_DEVICES"

# ---- host paths (device-plugin side) ----
MANAGER_DIR = "/etc/vgpu-manager"
DRIVER_LIB_NAME = "libvgpu-control.so"

# limits
MAX_DEVICE_COUNT = 16
CORES_PER_GPU = 100  # vgpu-cores granularity (% of one GPU)
