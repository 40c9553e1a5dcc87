"""Pod-annotation protocol — the scheduler-extender handshake, pod side.

Wire-compatible with the reference (pkg/gpu/nvidia/podutils.go): the
gpushare-scheduler-extender "assumes" a pod onto a GPU by writing
``ALIYUN_COM_GPU_MEM_IDX`` + ``ALIYUN_COM_GPU_MEM_ASSUME_TIME`` +
``ALIYUN_COM_GPU_MEM_ASSIGNED=false`` annotations; the plugin's Allocate
confirms the binding by patching ``ASSIGNED=true``.

Pods are plain k8s JSON dicts (what the kubelet /pods endpoint and the
apiserver return) — no typed client needed.
"""

from __future__ import annotations

import json
import time
from typing import Optional

from .. import consts


# --------------------------------------------------------------------------- #
# accessors
# --------------------------------------------------------------------------- #

def annotations(pod: dict) -> dict:
    # pods are untyped JSON from /pods or the apiserver — a malformed
    # payload must degrade to "no annotations", not an AttributeError in
    # every caller
    md = pod.get("metadata", {})
    anns = md.get("annotations") if isinstance(md, dict) else None
    return anns if isinstance(anns, dict) else {}


def pod_name(pod: dict) -> str:
    return pod.get("metadata", {}).get("name", "")


def pod_namespace(pod: dict) -> str:
    return pod.get("metadata", {}).get("namespace", "default")


def pod_uid(pod: dict) -> str:
    return pod.get("metadata", {}).get("uid", "")


def pod_phase(pod: dict) -> str:
    return pod.get("status", {}).get("phase", "")


def gpu_memory_of_pod(pod: dict) -> int:
    """Sum of `aliyun.com/gpu-mem` limits over containers
    (reference: getGPUMemoryFromPodResource, podutils.go:122-131)."""
    total = 0
    for c in pod.get("spec", {}).get("containers", []):
        limits = c.get("resources", {}).get("limits") or {}
        val = limits.get(consts.RESOURCE_NAME)
        if val is not None:
            total += int(val)
    return total


def gpu_memory_of_container(container: dict) -> int:
    limits = container.get("resources", {}).get("limits") or {}
    return int(limits.get(consts.RESOURCE_NAME, 0))


def gpu_id_from_annotation(pod: dict) -> int:
    """GPU index bound by the extender; -1 if absent/bad
    (reference: getGPUIDFromPodAnnotation, podutils.go:37-61)."""
    val = annotations(pod).get(consts.ENV_RESOURCE_INDEX)
    if val is None:
        return -1
    try:
        return int(val)
    except ValueError:
        return -1


def assume_time_from_annotation(pod: dict) -> int:
    """Extender's assume timestamp in ns; 0 if absent/bad
    (reference: getAssumeTimeFromPodAnnotation, podutils.go:64-76)."""
    val = annotations(pod).get(consts.ENV_RESOURCE_ASSUME_TIME)
    try:
        return int(val)
    except (TypeError, ValueError):
        return 0


def allocation_map_from_annotation(pod: dict) -> Optional[dict]:
    """Per-container allocation map from newer extenders:
    ``{container_name: {gpu_idx_str: mem_units}}``
    (reference: cmd/inspect/nodeinfo.go:244-271)."""
    raw = annotations(pod).get(consts.ANN_GPUSHARE_ALLOCATION)
    if not raw:
        return None
    try:
        parsed = json.loads(raw)
        return parsed if isinstance(parsed, dict) else None
    except json.JSONDecodeError:
        return None


def gpu_split_from_pod(pod: dict) -> Optional[dict[int, int]]:
    """Merged per-GPU placement {gpu_idx: units} from the allocation-map
    annotation (summed across containers); None if the annotation is
    absent/invalid.  A multi-GPU pod's authoritative placement record."""
    alloc = allocation_map_from_annotation(pod)
    if not alloc:
        return None
    merged: dict[int, int] = {}
    try:
        for per_gpu in alloc.values():
            for idx_s, units in per_gpu.items():
                idx, u = int(idx_s), int(units)
                # annotations are cluster data any pod-patcher can write:
                # reject absurd indices/units instead of letting them
                # poison the extender's resync ledger (1 Mi units = 1 PiB
                # at GiB grain — far beyond any node)
                if not (0 <= idx < 1024 and 0 < u <= 1 << 20):
                    return None
                merged[idx] = merged.get(idx, 0) + u
    except (TypeError, ValueError, AttributeError):
        return None
    return merged or None


# --------------------------------------------------------------------------- #
# predicates
# --------------------------------------------------------------------------- #

def is_assumed_pod(pod: dict) -> bool:
    """Pod assumed by the extender but not yet assigned by us
    (reference: isGPUMemoryAssumedPod, podutils.go:78-119 — requires a
    gpu-mem limit, an assume-time annotation, and ASSIGNED == "false")."""
    if gpu_memory_of_pod(pod) <= 0:
        return False
    anns = annotations(pod)
    if consts.ENV_RESOURCE_ASSUME_TIME not in anns:
        return False
    return anns.get(consts.ENV_ASSIGNED_FLAG) == "false"


def pod_is_not_running(pod: dict) -> bool:
    """Terminal-or-zombie predicate (reference: podIsNotRunning,
    podutils.go:133-147)."""
    meta = pod.get("metadata", {})
    status = pod.get("status", {})
    if meta.get("deletionTimestamp"):
        return True
    phase = status.get("phase", "")
    if phase in ("Failed", "Succeeded"):
        return True
    conditions = status.get("conditions") or []
    if phase == "Pending" and _condition_true_only(conditions, "PodScheduled"):
        return True
    return False


def _condition_true_only(conditions: list, expect: str) -> bool:
    if len(conditions) != 1:
        return False
    c = conditions[0]
    return c.get("type") == expect and c.get("status") == "True"


# --------------------------------------------------------------------------- #
# patches
# --------------------------------------------------------------------------- #

def assigned_patch(now_ns: Optional[int] = None) -> dict:
    """Strategic-merge patch marking the pod assigned (reference:
    patchPodAnnotationSpecAssigned, podutils.go:27-35 — note it refreshes
    ASSUME_TIME, which also re-sorts the pod out of the pending queue)."""
    if now_ns is None:
        now_ns = time.time_ns()
    return {
        "metadata": {
            "annotations": {
                consts.ENV_ASSIGNED_FLAG: "true",
                consts.ENV_RESOURCE_ASSUME_TIME: str(now_ns),
            }
        }
    }
