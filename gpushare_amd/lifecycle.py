"""Lifecycle manager — daemon run loop (reference: pkg/gpu/nvidia/gpumanager.go).

Responsibilities (reference parity, gpumanager.go:23-111):
- device-source init and device-count check (we fail loudly instead of the
  reference's silent forever-block on an empty select{}, gpumanager.go:44-47);
- node ``gpu-count`` patch + isolation-label check before serving;
- plugin construct → Serve (listen + self-dial + register);
- event loop: kubelet.sock re-creation (inotify) → full plugin restart,
  SIGHUP → restart, SIGQUIT → all-thread stack dump, SIGINT/SIGTERM →
  graceful stop;
- health monitor start/stop around the plugin (health.py).

Crash-only invariant preserved: all allocation state lives in pod
annotations + node status, so a restart (ours or kubelet's) loses nothing
(reference §3.5).
"""

from __future__ import annotations

import logging
import os
import queue
import signal
import threading
from dataclasses import dataclass

from . import consts
from .allocator import Allocator
from .cluster.events import EventRecorder
from .cluster.podmanager import PodManager
from .coredump import coredump
from .deviceplugin.server import GPUSharePlugin
from .health import HealthMonitor
from .watchers import FSWatcher, IN_CREATE

log = logging.getLogger(__name__)


@dataclass
class ManagerOptions:
    memory_unit: str = consts.GIB
    query_kubelet: bool = False
    health_check: bool = False
    deep_probe_interval: float = 0.0
    socket_dir: str = consts.DEVICE_PLUGIN_PATH
    cache_ttl: float = 0.2
    inject_devices: bool = True
    coredump_dir: str = "/etc/kubernetes"
    use_informer: bool = True    # watch-based pending-pod cache (informer.py);
                                 # falls back to LIST paths while unsynced
    memguard_path: str = ""      # host path of libgpushare_memguard.so;
                                 # "" disables VRAM budget enforcement
    allow_oversize_inventory: bool = False
    trace_file: str = ""         # JSONL per-Allocate trace (debugging)
    probe_mode: str = "subprocess"  # canary probes in a short-lived child
                                    # (daemon never maps the HIP runtime);
                                    # "inproc" keeps the r1 behavior
    numa_topology: bool = False  # advertise Device.topology NUMA hints
                                 # (modern kubelet TopologyManager field)


class SharedGPUManager:
    def __init__(
        self,
        source,
        kube_client,
        node_name: str,
        kubelet_client=None,
        options: ManagerOptions = None,
    ):
        self.source = source
        self.kube = kube_client
        self.kubelet = kubelet_client
        self.node_name = node_name
        self.opt = options or ManagerOptions()
        self.plugin: GPUSharePlugin = None
        self.health: HealthMonitor = None
        self.pod_informer = None
        self._signals: "queue.Queue[int]" = queue.Queue()
        self._stop = threading.Event()

    # ------------------------------------------------------------------ #
    def _build_plugin(self) -> GPUSharePlugin:
        gpus = self.source.devices()
        if not gpus:
            raise RuntimeError("no GPUs found; refusing to serve")
        if (
            self.opt.use_informer
            and self.pod_informer is None
            # in-memory fakes (tests, mock mode) have no watch stream
            and hasattr(self.kube, "watch_pods_stream")
        ):
            from .cluster.informer import PodInformer

            self.pod_informer = PodInformer(self.kube, self.node_name).start()
        pm = PodManager(
            self.kube,
            self.node_name,
            kubelet_client=self.kubelet,
            query_kubelet=self.opt.query_kubelet,
            cache_ttl=self.opt.cache_ttl,
            informer=self.pod_informer,
        )
        pm.patch_gpu_count(len(gpus))
        pm.patch_topology(gpus, unit=self.opt.memory_unit)
        allocator = Allocator(
            gpus,
            pm,
            unit=self.opt.memory_unit,
            disable_isolation=pm.isolation_disabled(),
            inject_devices=self.opt.inject_devices,
            event_recorder=EventRecorder(self.kube, self.node_name),
            memguard_path=self.opt.memguard_path,
            trace_file=self.opt.trace_file,
        )
        return GPUSharePlugin(
            gpus,
            allocator,
            unit=self.opt.memory_unit,
            socket_dir=self.opt.socket_dir,
            allow_oversize_inventory=self.opt.allow_oversize_inventory,
            numa_topology=self.opt.numa_topology,
        )

    def _start_plugin(self) -> None:
        self.plugin = self._build_plugin()
        self.plugin.serve()
        if self.opt.health_check or self.opt.deep_probe_interval > 0:
            self.health = HealthMonitor(
                self.source,
                self.plugin,
                deep_probe_interval=self.opt.deep_probe_interval,
                event_recorder=EventRecorder(self.kube, self.node_name),
                probe_mode=self.opt.probe_mode,
            )
            self.health.start()

    def _stop_plugin(self) -> None:
        if self.health is not None:
            self.health.stop()
            self.health = None
        if self.plugin is not None:
            self.plugin.stop()
            self.plugin = None

    def restart(self) -> None:
        log.info("restarting device plugin")
        self._stop_plugin()
        self._start_plugin()

    # ------------------------------------------------------------------ #
    def install_signal_handlers(self) -> None:
        for sig in (signal.SIGHUP, signal.SIGQUIT, signal.SIGINT, signal.SIGTERM):
            signal.signal(sig, lambda s, _f: self._signals.put(s))

    def run(self, max_iterations: int = 0) -> None:
        """Serve until SIGINT/SIGTERM.  ``max_iterations`` bounds the event
        loop for tests (0 = forever)."""
        os.makedirs(self.opt.socket_dir, exist_ok=True)
        watcher = FSWatcher()
        watcher.add(self.opt.socket_dir)
        self._start_plugin()
        iterations = 0
        try:
            while not self._stop.is_set():
                if max_iterations and iterations >= max_iterations:
                    break
                iterations += 1
                for _dir, name, mask in watcher.poll(timeout=0.5):
                    if name == consts.KUBELET_SOCKET_NAME and mask & IN_CREATE:
                        log.warning(
                            "inotify: %s created — kubelet restarted, "
                            "re-registering",
                            name,
                        )
                        self.restart()
                try:
                    sig = self._signals.get_nowait()
                except queue.Empty:
                    continue
                if sig == signal.SIGHUP:
                    log.info("SIGHUP: restarting")
                    self.restart()
                elif sig == signal.SIGQUIT:
                    path = coredump(self.opt.coredump_dir)
                    log.info("SIGQUIT: stacks dumped to %s", path)
                elif sig in (signal.SIGINT, signal.SIGTERM):
                    log.info("signal %d: shutting down", sig)
                    self._stop.set()
        finally:
            self._stop_plugin()
            if self.pod_informer is not None:
                self.pod_informer.stop()
                self.pod_informer = None
            watcher.close()

    def shutdown(self) -> None:
        self._stop.set()
