"""Wire-level constants: resource names, annotation schema, policies.

These are the compatibility contract with the elasticgpu.io ecosystem
(reference: pkg/utils/types.go:3-17 and the elasticgpu.io/elastic-gpu
v1alpha1 resource names used at pkg/scheduler/pod.go:28-30, 140-148), so a
user of the reference scheduler can switch without changing their pod specs.
"""
from __future__ import annotations

# --- Extended resource names (pod spec contract) ---
RESOURCE_GPU_CORE = "elasticgpu.io/gpu-core"
RESOURCE_GPU_MEMORY = "elasticgpu.io/gpu-memory"
RESOURCE_QGPU_CORE = "elasticgpu.io/qgpu-core"
RESOURCE_QGPU_MEMORY = "elasticgpu.io/qgpu-memory"
RESOURCE_PGPU = "elasticgpu.io/pgpu"

GPU_RESOURCE_NAMES = (
    RESOURCE_GPU_CORE,
    RESOURCE_GPU_MEMORY,
    RESOURCE_QGPU_CORE,
    RESOURCE_QGPU_MEMORY,
    RESOURCE_PGPU,
)

# The ROCm k8s device plugin's whole-card resource; nodes running the AMD
# device plugin advertise it. Our agent maps elasticgpu.io units onto it.
RESOURCE_AMD_GPU = "amd.com/gpu"

# One whole card == 100 gpu-core units (reference pkg/utils/types.go:6).
GPU_CORE_EACH_CARD = 100

# --- Annotation / label schema (contract with the node agent) ---
EGPU_ASSUMED = "elasticgpu.io/assumed"
ANNOTATION_EGPU_CONTAINER_PREFIX = "elasticgpu.io/container-"

# Per-pod placement constraint: "true" forces every container of the pod
# onto DISTINCT cards (the upstream README's "spread containers of pod to
# different GPUs" capability, which the reference never implements).
ANNOTATION_SPREAD_CONTAINERS = "elasticgpu.io/spread-containers"

# MI355X-native extensions (ours; absent on reference-scheduled pods):
ANNOTATION_EGPU_NODE = "elasticgpu.io/scheduled-node"
ANNOTATION_EGPU_SCORE = "elasticgpu.io/placement-score"
# Node-side annotations published by the MI355X agent:
ANNOTATION_NODE_TOPOLOGY = "elasticgpu.io/xgmi-topology"  # JSON hop matrix
ANNOTATION_NODE_INVENTORY = "elasticgpu.io/gpu-inventory"  # JSON per-card info

# --- Scheduling policies ---
PRIORITY_BINPACK = "binpack"
PRIORITY_SPREAD = "spread"
PRIORITY_RANDOM = "random"
PRIORITIES = (PRIORITY_BINPACK, PRIORITY_SPREAD, PRIORITY_RANDOM)

# --- Modes (reference cmd/main.go:29; only gpushare is live upstream) ---
MODE_GPUSHARE = "gpushare"
MODE_PGPU = "pgpu"
MODE_QGPU = "qgpu"
MODES = (MODE_GPUSHARE, MODE_PGPU, MODE_QGPU)

# Default HTTP port (reference cmd/main.go:69-72).
DEFAULT_PORT = 39999

# MI355X hardware defaults.
MI355X_MEMORY_BYTES = 288 * 1024**3
MI355X_CARDS_PER_NODE = 8
