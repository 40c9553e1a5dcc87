"""Extender core: cluster-state sync + the annotation-handshake writer.

Crash-only like the plugin: the ledger is soft state rebuilt from pod
annotations; the durable record of every placement is the
``ALIYUN_COM_GPU_MEM_IDX`` / ``ASSUME_TIME`` / ``ASSIGNED`` annotation set
this class writes (the exact triple the plugin's Allocate consumes,
reference: podutils.go:78-119).
"""

from __future__ import annotations

import json
import logging
import threading
import time
from typing import Optional

from .. import consts
from ..cluster import podutils
from .binpack import BinpackState

log = logging.getLogger(__name__)


class GPUShareExtender:
    def __init__(self, kube_client, resync_interval: float = 30.0,
                 policy: str = "binpack"):
        if policy not in ("binpack", "spread"):
            raise ValueError(f"unknown placement policy {policy!r}")
        self.kube = kube_client
        self.policy = policy
        self.state = BinpackState()
        self._lock = threading.Lock()
        self.resync_interval = resync_interval
        self._last_resync = 0.0
        self.assumed = 0
        self.rejected = 0
        # pods seen at filter time, reused at bind time (the scheduler always
        # filters before binding) — saves the bind-path GET to the apiserver
        self._pod_cache: dict[tuple, tuple] = {}
        self.pod_cache_ttl = 30.0
        # idempotence: the webhook `release` and the informer's DELETED
        # event may both fire for one pod — releasing twice would free
        # units another pod holds
        self._released: dict[str, float] = {}
        self.released_ttl = 10.0  # covers webhook+watch double-fire, keeps the table small
        self._last_released_prune = 0.0
        # placement record from assume time: release() falls back to it
        # when the caller's pod copy lacks the allocation-map annotation
        # (e.g. a delete-event stub); pruned on resync against live pods
        self._placements: dict[tuple, tuple] = {}   # (ns,name) -> (node, split)

    # ------------------------------------------------------------------ #
    # state sync
    # ------------------------------------------------------------------ #
    def register_node(self, node: str, per_gpu_units: list[int],
                      xgmi: Optional[list] = None,
                      numa: Optional[list] = None) -> None:
        self.state.set_node(node, per_gpu_units, xgmi=xgmi, numa=numa)

    def resync(self, nodes: Optional[list[str]] = None) -> None:
        """Rebuild the ledger from pod annotations (source of truth)."""
        nodes = nodes or list(self.state.nodes.keys())
        pods = self.kube.list_pods().get("items", [])
        for node in nodes:
            st = self.state.nodes.get(node)
            if st is None:
                continue
            allocated = [0] * len(st.per_gpu_units)
            for pod in pods:
                if pod.get("spec", {}).get("nodeName") != node:
                    continue
                if podutils.pod_is_not_running(pod):
                    continue
                mem = podutils.gpu_memory_of_pod(pod)
                if mem <= 0:
                    continue
                anns = podutils.annotations(pod)
                if consts.ENV_RESOURCE_ASSUME_TIME not in anns:
                    continue
                split = podutils.gpu_split_from_pod(pod)
                if split:
                    for idx, units in split.items():
                        if 0 <= idx < len(allocated):
                            allocated[idx] += units
                    continue
                idx = podutils.gpu_id_from_annotation(pod)
                if 0 <= idx < len(allocated):
                    allocated[idx] += mem
            self.state.set_node(
                node, st.per_gpu_units, allocated, st.xgmi, numa=st.numa
            )
        live = {
            (podutils.pod_namespace(p), podutils.pod_name(p)) for p in pods
        }
        with self._lock:
            self._placements = {
                k: v for k, v in self._placements.items() if k in live
            }
        self._last_resync = time.monotonic()

    def _maybe_resync(self) -> None:
        if time.monotonic() - self._last_resync > self.resync_interval:
            try:
                self.resync()
            except Exception as e:  # noqa: BLE001
                log.warning("extender resync failed: %s", e)

    # ------------------------------------------------------------------ #
    # scheduling verbs
    # ------------------------------------------------------------------ #
    def filter(self, pod: dict, node_names: list[str]) -> list[str]:
        """Webhook `filter`: nodes with a GPU that fits the pod."""
        self._maybe_resync()
        name = podutils.pod_name(pod)
        if name:
            key = (podutils.pod_namespace(pod), name)
            with self._lock:
                self._pod_cache[key] = (pod, time.monotonic())
                if len(self._pod_cache) > 10_000:
                    cutoff = time.monotonic() - self.pod_cache_ttl
                    self._pod_cache = {
                        k: v for k, v in self._pod_cache.items() if v[1] > cutoff
                    }
        request = podutils.gpu_memory_of_pod(pod)
        if request <= 0:
            return node_names
        return self.state.filter_nodes(request, node_names)

    def assume(self, pod: dict, node: str) -> Optional[int]:
        """Webhook `bind` body: best-fit GPU(s), write the annotation triple.
        A request no single GPU can hold is split over the smallest
        most-xGMI-connected set (binpack.best_fit_multi) and recorded in the
        per-container allocation-map annotation the plugin and inspect CLI
        read.  Returns the primary (lowest) GPU index, or None if the node
        cannot fit the pod."""
        request = podutils.gpu_memory_of_pod(pod)
        if request <= 0:
            return None
        split = self.state.assume_multi(
            node, request, spread=(self.policy == "spread")
        )
        if split is None:
            self.rejected += 1
            return None
        idx = min(split)
        ns = podutils.pod_namespace(pod)
        name = podutils.pod_name(pod)
        anns = {
            consts.ENV_RESOURCE_INDEX: str(idx),
            consts.ENV_RESOURCE_ASSUME_TIME: str(time.time_ns()),
            consts.ENV_ASSIGNED_FLAG: "false",
        }
        if len(split) > 1:
            # carve the split across the pod's GPU-requesting containers in
            # spec order (reference per-container map format,
            # cmd/inspect/nodeinfo.go:244-271): each container's per-GPU
            # units sum to its own gpu-mem limit, and per GPU the
            # containers sum to the split — so inspect/top attribution and
            # the plugin's per-device budget carve agree
            requesting = [
                (
                    c.get("name", f"c{i}"),
                    int(
                        c.get("resources", {})
                        .get("limits", {})
                        .get(consts.RESOURCE_NAME, 0)
                    ),
                )
                for i, c in enumerate(pod.get("spec", {}).get("containers", []))
            ]
            requesting = [(n, u) for n, u in requesting if u > 0]
            if not requesting:
                requesting = [("main", request)]
            remaining = dict(sorted(split.items()))
            alloc_map: dict[str, dict[str, int]] = {}
            for cname, units in requesting:
                need = units
                per_gpu: dict[str, int] = {}
                for gpu_idx in list(remaining):
                    if need <= 0:
                        break
                    take = min(remaining[gpu_idx], need)
                    if take > 0:
                        per_gpu[str(gpu_idx)] = take
                        remaining[gpu_idx] -= take
                        need -= take
                alloc_map[cname] = per_gpu
            anns[consts.ANN_GPUSHARE_ALLOCATION] = json.dumps(alloc_map)
        patch = {"metadata": {"annotations": anns}}
        try:
            self.kube.patch_pod(ns, name, patch, parse=False)
        except Exception as e:  # noqa: BLE001
            log.warning("assume patch failed for %s/%s: %s", ns, name, e)
            self.state.release_multi(node, split)
            return None
        with self._lock:
            self._placements[(ns, name)] = (node, split)
        self.assumed += 1
        return idx

    def release(self, pod: dict, node: str) -> None:
        """Return the pod's reservation (webhook `release` or the
        informer's DELETED event — idempotent across both)."""
        uid = podutils.pod_uid(pod)
        if uid:  # uid-less caller stubs are always honored
            now = time.monotonic()
            with self._lock:
                if self._released.get(uid, 0.0) > now:
                    return
                self._released[uid] = now + self.released_ttl
                # amortized: never an O(n) rebuild per release
                if (
                    len(self._released) > 10_000
                    and now - self._last_released_prune > 1.0
                ):
                    self._released = {
                        u: t for u, t in self._released.items() if t > now
                    }
                    self._last_released_prune = now
        key = (podutils.pod_namespace(pod), podutils.pod_name(pod))
        with self._lock:
            recorded = self._placements.pop(key, None)
        split = podutils.gpu_split_from_pod(pod)
        if split is None and recorded is not None:
            node, split = recorded[0] or node, recorded[1]
        if split:
            self.state.release_multi(node, split)
            return
        request = podutils.gpu_memory_of_pod(pod)
        idx = podutils.gpu_id_from_annotation(pod)
        if request > 0 and idx >= 0:
            self.state.release(node, idx, request)

    def cached_pod(self, namespace: str, name: str):
        """Pod captured at filter time, if still fresh (one-shot)."""
        with self._lock:
            entry = self._pod_cache.pop((namespace, name), None)
        if entry is None:
            return None
        pod, t = entry
        if time.monotonic() - t > self.pod_cache_ttl:
            return None
        return pod

    def prioritize(self, pod: dict, node_names: list[str]) -> list[dict]:
        """Webhook `prioritize`: HostPriorityList scoring feasible nodes by
        binpack tightness (k8s extender max score 10).  The reference
        extender exposes filter+bind only; without scoring the default
        spreading policy works against co-location."""
        request = podutils.gpu_memory_of_pod(pod)
        if request <= 0:
            return [{"Host": n, "Score": 0} for n in node_names]
        scores = self.state.score_nodes(
            request, node_names, spread=(self.policy == "spread")
        )
        return [
            {"Host": n, "Score": scores.get(n, 0)} for n in node_names
        ]

    def packing(self) -> dict:
        return self.state.packing()


def make_auto_release_informer(kube, extender: GPUShareExtender):
    """All-nodes pod watch that returns a deleted gpushare pod's
    reservation immediately (production flow: kubectl delete → watch
    DELETED → release; idempotent with the webhook release path).  The
    caller starts/stops the returned informer."""
    from ..cluster.informer import PodInformer

    def _on_event(etype: str, pod: dict) -> None:
        if etype != "DELETED" or podutils.gpu_memory_of_pod(pod) <= 0:
            return
        node = pod.get("spec", {}).get("nodeName", "")
        if node:
            extender.release(pod, node)

    return PodInformer(kube, None, on_event=_on_event)
