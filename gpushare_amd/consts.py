"""Wire-contract constants.

The resource names, socket names, and the ``ALIYUN_COM_GPU_MEM_*`` pod
annotation/env protocol are external contracts shared with the
gpushare-scheduler-extender and existing workloads; they are kept
byte-for-byte compatible with the reference
(reference: pkg/gpu/nvidia/const.go:10-36).  Everything device-side is
AMD-native.
"""

# ---------------------------------------------------------------------------
# Kubelet device-plugin API v1beta1 contract
# (reference: vendor/k8s.io/kubernetes/pkg/kubelet/apis/deviceplugin/v1beta1/constants.go)
# ---------------------------------------------------------------------------
HEALTHY = "Healthy"
UNHEALTHY = "Unhealthy"
API_VERSION = "v1beta1"
DEVICE_PLUGIN_PATH = "/var/lib/kubelet/device-plugins/"
KUBELET_SOCKET_NAME = "kubelet.sock"

# Our plugin's own unix socket name inside DEVICE_PLUGIN_PATH.
# (reference: const.go:13 "aliyungpushare.sock"; ours is the AMD daemon)
SERVER_SOCK_NAME = "amdgpushare.sock"

# Max length of a Device.ID on the wire (api.proto:84).
MAX_DEVICE_ID_LEN = 63

# The kubelet's gRPC client receives with the Go default 4 MiB cap; a
# ListAndWatchResponse above it is silently fatal (RESOURCE_EXHAUSTED on
# the kubelet side).  At MiB grain one 288 GiB MI355X already encodes to
# ~8.4 MB — see GPUSharePlugin's inventory guard.
KUBELET_GRPC_MAX_MSG = 4 << 20

# ---------------------------------------------------------------------------
# Extended resources (scheduler contract — unchanged from the reference so the
# gpushare-scheduler-extender works as-is; reference: const.go:11-12)
# ---------------------------------------------------------------------------
RESOURCE_NAME = "aliyun.com/gpu-mem"
RESOURCE_COUNT = "aliyun.com/gpu-count"

# ---------------------------------------------------------------------------
# Pod annotation / env protocol (scheduler-extender handshake,
# reference: const.go:25-31 + podutils.go)
# ---------------------------------------------------------------------------
ENV_RESOURCE_INDEX = "ALIYUN_COM_GPU_MEM_IDX"
ENV_RESOURCE_BY_POD = "ALIYUN_COM_GPU_MEM_POD"
ENV_RESOURCE_BY_CONTAINER = "ALIYUN_COM_GPU_MEM_CONTAINER"
ENV_RESOURCE_BY_DEV = "ALIYUN_COM_GPU_MEM_DEV"
ENV_ASSIGNED_FLAG = "ALIYUN_COM_GPU_MEM_ASSIGNED"
ENV_RESOURCE_ASSUME_TIME = "ALIYUN_COM_GPU_MEM_ASSUME_TIME"
ENV_RESOURCE_ASSIGN_TIME = "ALIYUN_COM_GPU_MEM_ASSIGN_TIME"

# Per-container allocation map written by newer scheduler-extender versions
# (reference: cmd/inspect/nodeinfo.go:244-271): JSON {container: {gpuIdx: mem}}.
# Also the record of a multi-GPU placement (a pod whose gpu-mem request spans
# several xGMI-linked MI355X) — the inspect CLI and the plugin's Allocate
# both read it.
ANN_GPUSHARE_ALLOCATION = "scheduler.framework.gpushare.allocation"

# Node annotation published by the plugin: real per-GPU capacities and the
# xGMI adjacency from KFD topology, JSON
# {"unit": "GiB", "per_gpu_units": [..], "xgmi": [[peer idx..], ..]}.
# The scheduler extender uses it to (a) place multi-GPU pods on
# link-adjacent GPU sets so RCCL inside the containers runs over xGMI, and
# (b) drop the homogeneous-capacity assumption the reference bakes in
# (nvidia.go:70-72 reads GPU0's memory for every device).
ANN_NODE_TOPOLOGY = "gpushare.amd.com/topology"

# ---------------------------------------------------------------------------
# AMD-native container injection (replaces NVIDIA_VISIBLE_DEVICES; on ROCm the
# env alone does nothing — the /dev/kfd + /dev/dri render nodes must be
# injected as DeviceSpec entries, which Allocate() does).
# ---------------------------------------------------------------------------
ENV_HIP_VISIBLE = "HIP_VISIBLE_DEVICES"
ENV_ROCR_VISIBLE = "ROCR_VISIBLE_DEVICES"

DEV_KFD = "/dev/kfd"
DEV_DRI_DIR = "/dev/dri"

# LD_PRELOAD VRAM budget enforcer (native/memguard.cpp) — the MI355X-native
# replacement for the reference's closed-source cGPU isolation module.
# Allocate() mounts the library read-only at MEMGUARD_CONTAINER_PATH and
# sets ENV_MEMGUARD_LIMIT to the container's gpu-mem share in bytes.
ENV_MEMGUARD_LIMIT = "GPUSHARE_MEM_LIMIT_BYTES"
ENV_MEMGUARD_POD_UID = "GPUSHARE_POD_UID"
# Per-container token scoping the /dev/shm accounting table: containers of
# one pod share /dev/shm but not a PID namespace, so liveness sweeps must
# never see a sibling container's pids.  The kubelet batches
# every gpu-mem container of a pod into one Allocate call, so the container
# index within the request is unique per pod.
ENV_MEMGUARD_CONTAINER_TOKEN = "GPUSHARE_CONTAINER_TOKEN"
# Comma-separated per-visible-ordinal byte caps (aligned with the injected
# HIP_VISIBLE_DEVICES order) enforcing the extender's per-GPU split on
# multi-GPU placements.
ENV_MEMGUARD_PER_DEVICE = "GPUSHARE_MEM_LIMIT_BYTES_PER_DEVICE"
MEMGUARD_CONTAINER_PATH = "/usr/local/lib/gpushare/libgpushare_memguard.so"

# Node label opting a node out of kernel-level isolation (analogue of the
# reference's cgpu.disable.isolation toggle, podmanager.go:59-72).
LABEL_DISABLE_ISOLATION = "cgpu.disable.isolation"
ENV_CGPU_DISABLE = "CGPU_DISABLE"

# ---------------------------------------------------------------------------
# Failure-path contract: Allocate() failures return a *successful* RPC whose
# envs poison the pod visibly (reference: allocate.go:24-39).  String format
# preserved: "no-gpu-has-<N><unit>-to-run".
# ---------------------------------------------------------------------------
def poisoned_visible_devices(req_units: int, unit: str) -> str:
    return f"no-gpu-has-{req_units}{unit}-to-run"


# Optimistic-concurrency conflict detection for annotation patches
# (reference: const.go:15; we match on HTTP 409 primarily, this string as a
# fallback for proxies that rewrite status codes).
OPTIMISTIC_LOCK_ERROR_MSG = (
    "the object has been modified; please apply your changes to the latest "
    "version and try again"
)

# Memory units (reference: const.go:34-35).
GIB = "GiB"
MIB = "MiB"
VALID_MEMORY_UNITS = (GIB, MIB)
