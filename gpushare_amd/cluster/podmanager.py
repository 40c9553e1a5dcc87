"""Pending-pod listing, candidate selection, and node patching.

Reference behavior preserved (pkg/gpu/nvidia/podmanager.go):
- primary path: kubelet read-only /pods (fast, node-local), with retries and
  an apiserver fallback (podmanager.go:124-139);
- fallback path: apiserver LIST with fieldSelector
  ``spec.nodeName=<node>,status.phase=Pending`` (podmanager.go:141-159);
- dedup by UID, node filter, assumed-pod filter, FIFO sort by assume time
  (podmanager.go:162-262).

MI355X-first delta (SURVEY §7 hard part 5): the reference performs 1–2
remote list calls inside *every* Allocate under a global mutex — that is its
p99 floor.  Here a small TTL cache fronts the listing; an Allocate that
fails to match in the cached view forces one revalidating refresh before
giving up, so correctness never depends on the TTL.  All durable state stays
in pod annotations/node status (crash-only, reference gpumanager design).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Optional

from .. import consts
from . import podutils
from .kubeclient import KubeError

log = logging.getLogger(__name__)

# Reference retry budgets (podmanager.go:26, 127-131, 148-154).  Tunable —
# the bench configs measure their effect on Allocate p99.
KUBELET_RETRIES = 8
KUBELET_RETRY_INTERVAL = 0.1
APISERVER_RETRIES = 3
APISERVER_RETRY_INTERVAL = 1.0


class PodManager:
    def __init__(
        self,
        kube_client,
        node_name: str,
        kubelet_client=None,
        query_kubelet: bool = True,
        cache_ttl: float = 0.2,
        kubelet_retries: int = KUBELET_RETRIES,
        kubelet_retry_interval: float = KUBELET_RETRY_INTERVAL,
        apiserver_retries: int = APISERVER_RETRIES,
        apiserver_retry_interval: float = APISERVER_RETRY_INTERVAL,
        informer=None,
    ):
        self.kube = kube_client
        self.kubelet = kubelet_client
        self.informer = informer      # cluster.informer.PodInformer (optional)
        self.node_name = node_name
        self.query_kubelet = query_kubelet and kubelet_client is not None
        self.cache_ttl = cache_ttl
        self.kubelet_retries = kubelet_retries
        self.kubelet_retry_interval = kubelet_retry_interval
        self.apiserver_retries = apiserver_retries
        self.apiserver_retry_interval = apiserver_retry_interval
        self._cache_lock = threading.Condition()
        self._cached_pods: Optional[list] = None
        self._cache_time = 0.0          # when the cached list STARTED
        self._refreshing = False

    # ------------------------------------------------------------------ #
    # listing
    # ------------------------------------------------------------------ #
    def _list_via_kubelet(self) -> list:
        last_err: Optional[Exception] = None
        for attempt in range(self.kubelet_retries + 1):
            try:
                podlist = self.kubelet.get_node_running_pods()
                pending = [
                    p
                    for p in podlist.get("items", [])
                    if podutils.pod_phase(p) == "Pending"
                ]
                if pending:
                    return pending
                last_err = KubeError(404, "not found pending pod")
            except (KubeError, OSError) as e:
                last_err = e
            if attempt < self.kubelet_retries:
                time.sleep(self.kubelet_retry_interval)
        raise last_err  # type: ignore[misc]

    def _list_via_apiserver(self) -> list:
        selector = f"spec.nodeName={self.node_name},status.phase=Pending"
        last_err: Optional[Exception] = None
        for attempt in range(self.apiserver_retries + 1):
            try:
                return self.kube.list_pods(field_selector=selector).get("items", [])
            except (KubeError, OSError) as e:
                last_err = e
            if attempt < self.apiserver_retries:
                time.sleep(self.apiserver_retry_interval)
        raise RuntimeError(
            f"failed to get Pods assigned to node {self.node_name}: {last_err}"
        )

    def get_pending_pods(
        self, force_refresh: bool = False, authoritative: bool = False
    ) -> list:
        """Pending pods on this node, deduped by UID (reference:
        getPendingPodsInNode, podmanager.go:162-212).

        Fast path: the informer's in-memory store — zero remote calls
        (watch events land push-style; the store is already node-scoped
        and uid-keyed).  ``authoritative=True`` bypasses the informer for
        a direct remote list — the allocator's final matching attempt uses
        it so scheduling never depends on watch freshness.

        Remote lists are single-flighted: N concurrent callers (N parallel
        Allocates missing the cache at once) trigger ONE remote list; a
        forced caller is satisfied only by a list that *started after it
        asked* (a list already in flight may predate the caller's own
        bind)."""
        if (
            not authoritative
            and self.informer is not None
            and self.informer.synced
        ):
            return self.informer.pending_pods()
        entered = time.monotonic()
        with self._cache_lock:
            while True:
                if self._cached_pods is not None:
                    if force_refresh:
                        if self._cache_time >= entered:
                            return self._cached_pods
                    elif entered - self._cache_time < self.cache_ttl:
                        return self._cached_pods
                if not self._refreshing:
                    self._refreshing = True
                    break
                self._cache_cond_wait()

        started = time.monotonic()
        result = None
        try:
            if self.query_kubelet:
                try:
                    pods = self._list_via_kubelet()
                except (KubeError, OSError):
                    log.warning(
                        "kubelet /pods failed after retries; "
                        "falling back to apiserver"
                    )
                    pods = self._list_via_apiserver()
            else:
                pods = self._list_via_apiserver()

            seen: set = set()
            result = []
            for pod in pods:
                if pod.get("spec", {}).get("nodeName") != self.node_name:
                    continue
                uid = podutils.pod_uid(pod)
                if uid in seen:
                    continue
                seen.add(uid)
                result.append(pod)
            return result
        finally:
            with self._cache_lock:
                self._refreshing = False
                if result is not None:
                    self._cached_pods = result
                    self._cache_time = started
                self._cache_lock.notify_all()

    def _cache_cond_wait(self) -> None:
        self._cache_lock.wait(timeout=5.0)

    def invalidate_cache(self) -> None:
        with self._cache_lock:
            self._cached_pods = None

    def _mark_cached_assigned(self, pod: dict) -> None:
        """Coherence-only cache update after a successful ASSIGNED patch —
        keeps steady-state Allocate at zero remote list calls (durable state
        is the patch itself; the cache refresh would rediscover this)."""
        uid = podutils.pod_uid(pod)
        with self._cache_lock:
            if self._cached_pods is None:
                return
            for p in self._cached_pods:
                if podutils.pod_uid(p) == uid:
                    p.setdefault("metadata", {}).setdefault("annotations", {})[
                        consts.ENV_ASSIGNED_FLAG
                    ] = "true"

    def get_candidate_pods(
        self, force_refresh: bool = False, authoritative: bool = False
    ) -> list:
        """Assumed-but-unassigned pods, oldest assume-time first (FIFO
        fairness; reference: getCandidatePods, podmanager.go:215-262)."""
        pods = self.get_pending_pods(
            force_refresh=force_refresh, authoritative=authoritative
        )
        candidates = [p for p in pods if podutils.is_assumed_pod(p)]
        candidates.sort(key=podutils.assume_time_from_annotation)
        return candidates

    # ------------------------------------------------------------------ #
    # pod + node patches
    # ------------------------------------------------------------------ #
    def mark_assigned(self, pod: dict, retries: int = 1) -> bool:
        """PATCH ASSIGNED=true with retry on optimistic-lock conflict
        (reference: allocate.go:131-149 — one retry)."""
        from .kubeclient import ConflictError

        ns, name = podutils.pod_namespace(pod), podutils.pod_name(pod)
        patch = podutils.assigned_patch()
        for attempt in range(retries + 1):
            try:
                self.kube.patch_pod(ns, name, patch, parse=False)
                self._mark_cached_assigned(pod)
                return True
            except ConflictError:
                if attempt < retries:
                    log.info("conflict patching pod %s/%s, retrying", ns, name)
                    continue
                log.error("giving up patching pod %s/%s after conflict", ns, name)
            except KubeError as e:
                log.error("failed to patch pod %s/%s: %s", ns, name, e)
                break
        return False

    def patch_gpu_count(self, gpu_count: int) -> None:
        """Write `aliyun.com/gpu-count` into node capacity+allocatable
        (reference: patchGPUCount, podmanager.go:74-99)."""
        node = self.kube.get_node(self.node_name)
        cap = node.get("status", {}).get("capacity", {})
        if cap.get(consts.RESOURCE_COUNT) == str(gpu_count):
            log.info("gpu count %d already patched", gpu_count)
            return
        patch = {
            "status": {
                "capacity": {consts.RESOURCE_COUNT: str(gpu_count)},
                "allocatable": {consts.RESOURCE_COUNT: str(gpu_count)},
            }
        }
        self.kube.patch_node_status(self.node_name, patch)

    def patch_topology(self, gpus: list, unit: str = consts.GIB) -> None:
        """Publish real per-GPU capacities + xGMI adjacency as a node
        annotation (MI355X-native: lets the extender place multi-GPU pods
        on link-adjacent sets and drop the reference's homogeneous-node
        assumption, nvidia.go:70-72)."""
        import json as _json

        topo = {
            "unit": unit,
            "per_gpu_units": [g.mem_units(unit) for g in gpus],
            "xgmi": [sorted(g.xgmi_peers) for g in gpus],
            "numa": [getattr(g, "numa_node", -1) for g in gpus],
        }
        patch = {
            "metadata": {
                "annotations": {consts.ANN_NODE_TOPOLOGY: _json.dumps(topo)}
            }
        }
        try:
            self.kube.patch_node(self.node_name, patch)
        except KubeError as e:
            log.warning("failed to publish topology annotation: %s", e)

    def isolation_disabled(self) -> bool:
        """Node label `cgpu.disable.isolation=true` check (reference:
        disableCGPUIsolationOrNot, podmanager.go:59-72)."""
        try:
            node = self.kube.get_node(self.node_name)
        except KubeError as e:
            log.warning("cannot read node %s: %s", self.node_name, e)
            return False
        labels = node.get("metadata", {}).get("labels") or {}
        return labels.get(consts.LABEL_DISABLE_ISOLATION, "").lower() == "true"
