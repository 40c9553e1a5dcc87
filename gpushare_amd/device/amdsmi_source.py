"""Real-GPU device source: in-tree ``_amdsmi`` dlopen shim + KFD topology.

AMD-native replacement for the reference's NVML layer
(pkg/gpu/nvidia/nvidia.go): enumeration and VRAM totals come from amdsmi;
render/card device nodes, NUMA affinity and xGMI peers come from the KFD
sysfs topology (matched by ``kfd_id``).

Health watching (reference analogue: watchXIDs, nvidia.go:100-152):
- passive amdsmi event notifications — thermal throttle and GPU pre/post
  reset flip health down/up; VM faults are per-process application errors
  and are ignored (the way the reference skips Xids 31/43/45, nvidia.go:134);
- an uncorrectable-ECC counter sweep every poll cycle (RAS), which NVML XIDs
  only expose indirectly.

Unlike the reference (Unhealthy is terminal — server.go:180 FIXME), a
``GPU_POST_RESET`` event restores the device to Healthy.
"""

from __future__ import annotations

import dataclasses
import logging
import threading
import time
from typing import Iterable

from . import HealthEvent, PhysicalGPU
from . import kfd_topology

log = logging.getLogger(__name__)

_EVENT_KINDS = {
    1: "vmfault",
    2: "thermal_throttle",
    3: "gpu_pre_reset",
    4: "gpu_post_reset",
}


def _load_shim():
    try:
        from .. import _amdsmi
    except ImportError as e:  # never fall back silently on a GPU box
        raise RuntimeError(
            "gpushare_amd._amdsmi native extension is not built; run "
            "`python -m gpushare_amd.native.build` (required for real-GPU "
            "mode; set GPUSHARE_MOCK_SPEC for mock mode)"
        ) from e
    return _amdsmi


class AmdSmiSource:
    def __init__(
        self,
        ecc_poll_interval: float = 30.0,
        topology_root: str = kfd_topology.KFD_TOPOLOGY,
        drm_root: str = kfd_topology.DRM_CLASS,
    ):
        self._smi = _load_shim()
        if not self._smi.available():
            raise RuntimeError(
                "libamd_smi.so not found — no ROCm runtime on this host "
                "(set GPUSHARE_MOCK_SPEC for mock mode)"
            )
        self._smi.init()
        self._ecc_poll_interval = ecc_poll_interval
        self._topology_root = topology_root
        self._drm_root = drm_root
        self._gpus = self._enumerate()
        self._lock = threading.Lock()

    def _enumerate(self) -> list[PhysicalGPU]:
        count = self._smi.device_count()
        if count == 0:
            raise RuntimeError("amdsmi reports 0 AMD GPUs on this node")
        topo = kfd_topology.resolve(self._topology_root, self._drm_root)
        gpus: list[PhysicalGPU] = []
        kfd_of_index: dict[int, int] = {}
        for i in range(count):
            info = self._smi.device_info(i)
            kfd_id = info.get("kfd_id")
            t = topo.get(kfd_id) if kfd_id is not None else None
            uuid = info.get("uuid") or None
            if not uuid:
                serial = (info.get("asic_serial") or "").lstrip("0x") or None
                uuid = f"amd-{serial}" if serial else f"amd-{info.get('bdf', i)}"
            # fake IDs must fit 63 chars with "-_-<grain>" appended
            uuid = uuid[:52]
            mem = int(info["vram_total_bytes"])
            if mem == 0 and t is not None:
                mem = t.vram_bytes
            kfd_of_index[i] = kfd_id
            gpus.append(
                PhysicalGPU(
                    index=i,
                    uuid=uuid,
                    memory_bytes=mem,
                    render_path=t.render_path if t else None,
                    card_path=t.card_path if t else None,
                    bdf=info.get("bdf"),
                    kfd_node=t.kfd_node_id if t else None,
                    numa_node=t.numa_node if t else -1,
                    xgmi_peers=(),  # filled below once all kfd ids known
                    extras={
                        "rocr_uuid": (
                            f"GPU-{t.unique_id:016x}" if t and t.unique_id else None
                        ),
                        "market_name": info.get("market_name"),
                        "num_compute_units": info.get("num_compute_units"),
                        "target_graphics_version": info.get(
                            "target_graphics_version"
                        ),
                    },
                )
            )
        # second pass: xGMI peers as plugin indices
        index_of_kfd = {v: k for k, v in kfd_of_index.items() if v is not None}
        resolved = []
        for g in gpus:
            t = topo.get(kfd_of_index.get(g.index))
            peers = tuple(
                sorted(
                    index_of_kfd[p]
                    for p in (t.xgmi_peer_gpu_ids if t else [])
                    if p in index_of_kfd
                )
            )
            resolved.append(dataclasses.replace(g, xgmi_peers=peers))
        return resolved

    def devices(self) -> list[PhysicalGPU]:
        return list(self._gpus)

    def process_usage(self) -> dict[int, list]:
        """Per-process VRAM/engine usage per GPU (amdsmi process list;
        includes container_name where the driver resolves the cgroup) —
        pod-level attribution for observability/debugging.  Note: pids are
        HOST-namespace (map to pods via /proc/<pid>/cgroup on the host)."""
        out: dict[int, list] = {}
        for g in self._gpus:
            try:
                out[g.index] = self._smi.process_list(g.index)
            except RuntimeError:
                out[g.index] = []
        return out

    def vram_usage(self) -> dict[int, int]:
        """Live per-GPU VRAM used bytes (observability; the scheduling
        currency stays the *allocated* annotations, not live usage)."""
        out: dict[int, int] = {}
        for g in self._gpus:
            try:
                info = self._smi.device_info(g.index)
            except RuntimeError:
                continue
            used = info.get("vram_used_bytes")
            if used is not None:
                out[g.index] = int(used)
        return out

    # ------------------------------------------------------------------ #
    def watch_health(self, stop_event) -> Iterable[HealthEvent]:
        smi = self._smi
        mask = smi.event_mask(
            [smi.EVT_THERMAL_THROTTLE, smi.EVT_GPU_PRE_RESET, smi.EVT_GPU_POST_RESET]
        )
        watched = []
        for g in self._gpus:
            try:
                smi.event_watch_init(g.index, mask)
                watched.append(g.index)
            except RuntimeError as e:
                log.warning("event watch unavailable for GPU %d: %s", g.index, e)
        last_ecc = {}
        last_ecc_check = 0.0
        try:
            while not stop_event.is_set():
                if watched:
                    try:
                        events = smi.event_poll(1000, 64)
                    except RuntimeError as e:
                        log.warning("event poll failed: %s", e)
                        events = []
                        stop_event.wait(1.0)
                else:
                    events = []
                    stop_event.wait(1.0)
                for idx, etype, msg in events:
                    kind = _EVENT_KINDS.get(etype, f"event_{etype}")
                    if etype == 1:  # vmfault: application-level, ignore
                        continue
                    healthy = etype == 4  # post_reset ⇒ recovered
                    yield HealthEvent(
                        gpu_index=None if idx < 0 else idx,
                        healthy=healthy,
                        kind=kind,
                        message=msg,
                    )
                now = time.monotonic()
                if now - last_ecc_check >= self._ecc_poll_interval:
                    last_ecc_check = now
                    for g in self._gpus:
                        try:
                            _, uncorr = smi.ecc_count(g.index)
                        except RuntimeError:
                            continue
                        prev = last_ecc.get(g.index)
                        last_ecc[g.index] = uncorr
                        if prev is not None and uncorr > prev:
                            yield HealthEvent(
                                gpu_index=g.index,
                                healthy=False,
                                kind="ecc_uncorrectable",
                                message=f"uncorrectable ECC count {prev}->{uncorr}",
                            )
        finally:
            for i in watched:
                try:
                    smi.event_watch_stop(i)
                except RuntimeError:
                    pass

    def close(self) -> None:
        try:
            self._smi.shutdown()
        except RuntimeError:
            pass
