"""Health monitoring — passive amdsmi events + optional active canary probes.

Reference analogue: the ``healthcheck()`` goroutine bridging watchXIDs to
the plugin's health channel (server.go:203-221, nvidia.go:100-152), with two
MI355X-native upgrades:

- a physical-GPU event flips **all** of that GPU's fake devices (plugin
  handles the fan-out; reference flipped one);
- an optional *active* probe launches the gfx950 canary kernels
  (``_canary.probe``: MFMA bit-exactness + VRAM pattern walk) on every GPU
  at a configurable interval — catching the shared-GPU failure mode a
  passive watcher cannot see: a device that still enumerates but no longer
  computes correctly.  Recovery flips the device back to Healthy (reference
  FIXME server.go:180: Unhealthy was terminal).
"""

from __future__ import annotations

import json
import logging
import os
import subprocess
import sys
import threading

log = logging.getLogger(__name__)


def probe_in_subprocess(
    gpu_index: int, vram_probe_mb: int = 32, timeout: float = 60.0
) -> dict:
    """Run one canary probe in a short-lived child process.

    The daemon itself must never map the HIP runtime: keeping
    libamdhip64 (plus its device-code images) resident cost ~10× the
    control plane's RSS in round 1 (~525 MB vs the reference's 300 Mi
    pod, profiles/perf_evolution_gpu_box.md) — and a canary that wedges
    the GPU now kills a disposable child, not the plugin.
    """
    import gpushare_amd

    code = (
        "import json, sys\n"
        "import gpushare_amd._canary as canary\n"
        "r = canary.probe(int(sys.argv[1]), vram_probe_mb=int(sys.argv[2]),"
        " bandwidth=False)\n"
        "print(json.dumps(r))\n"
    )
    env = dict(os.environ)
    pkg_root = os.path.dirname(os.path.dirname(gpushare_amd.__file__))
    env["PYTHONPATH"] = pkg_root + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.run(
        [sys.executable, "-c", code, str(gpu_index), str(vram_probe_mb)],
        capture_output=True,
        text=True,
        timeout=timeout,
        env=env,
    )
    if proc.returncode != 0:
        raise RuntimeError(
            f"canary subprocess rc={proc.returncode}: "
            f"{proc.stderr.strip()[-500:]}"
        )
    return json.loads(proc.stdout.strip().splitlines()[-1])


class HealthMonitor:
    def __init__(
        self,
        source,
        plugin,
        deep_probe_interval: float = 0.0,   # 0 = passive only
        probe_vram_mb: int = 32,
        event_recorder=None,
        probe_mode: str = "subprocess",     # "subprocess" | "inproc"
    ):
        self.source = source
        self.plugin = plugin
        self.deep_probe_interval = deep_probe_interval
        self.probe_vram_mb = probe_vram_mb
        self.events = event_recorder
        self.probe_mode = probe_mode
        self._stop = threading.Event()
        self._threads: list[threading.Thread] = []
        self._probe_failed: set[int] = set()

    # ------------------------------------------------------------------ #
    def start(self) -> None:
        self._stop.clear()
        t = threading.Thread(target=self._watch_passive, name="health-passive",
                             daemon=True)
        t.start()
        self._threads = [t]
        if self.deep_probe_interval > 0:
            t2 = threading.Thread(target=self._probe_loop, name="health-probe",
                                  daemon=True)
            t2.start()
            self._threads.append(t2)

    def stop(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=5)
        self._threads = []

    # ------------------------------------------------------------------ #
    def _watch_passive(self) -> None:
        try:
            for ev in self.source.watch_health(self._stop):
                log.warning(
                    "health event: gpu=%s healthy=%s kind=%s %s",
                    ev.gpu_index, ev.healthy, ev.kind, ev.message,
                )
                self.plugin.set_gpu_health(ev.gpu_index, ev.healthy)
                if self.events is not None:
                    self.events.emit(
                        "GPUHealthy" if ev.healthy else "GPUUnhealthy",
                        f"GPU {ev.gpu_index}: {ev.kind} {ev.message}".strip(),
                        etype="Normal" if ev.healthy else "Warning",
                    )
        except Exception as e:  # noqa: BLE001
            log.error("passive health watcher died: %s", e)

    def _probe_once(self, gpu_index: int) -> dict:
        if self.probe_mode == "inproc":
            import gpushare_amd._canary as canary

            return canary.probe(
                gpu_index, vram_probe_mb=self.probe_vram_mb, bandwidth=False
            )
        return probe_in_subprocess(gpu_index, self.probe_vram_mb)

    def _probe_loop(self) -> None:
        if self.probe_mode == "inproc":
            try:
                import gpushare_amd._canary  # noqa: F401
            except ImportError as e:
                log.error("deep probe requested but _canary not built: %s", e)
                return
        from . import metrics

        while not self._stop.wait(self.deep_probe_interval):
            if hasattr(self.source, "vram_usage"):
                try:
                    metrics.observe_vram_usage(self.source.vram_usage())
                except Exception as e:  # noqa: BLE001
                    log.warning("vram usage poll failed: %s", e)
            for gpu in self.plugin.gpus:
                if self._stop.is_set():
                    return
                try:
                    result = self._probe_once(gpu.index)
                    ok = bool(result.get("ok"))
                except (
                    RuntimeError,
                    subprocess.TimeoutExpired,
                    ValueError,
                ) as e:
                    log.error("canary probe GPU %d errored: %s", gpu.index, e)
                    ok = False
                    result = {"error": str(e)}
                if not ok and gpu.index not in self._probe_failed:
                    log.error("deep probe FAILED on GPU %d: %s", gpu.index, result)
                    self._probe_failed.add(gpu.index)
                    self.plugin.set_gpu_health(gpu.index, False)
                    if self.events is not None:
                        self.events.emit(
                            "GPUCanaryFailed",
                            f"GPU {gpu.index} deep probe failed: {result}",
                        )
                elif ok and gpu.index in self._probe_failed:
                    log.warning("deep probe recovered on GPU %d", gpu.index)
                    self._probe_failed.discard(gpu.index)
                    self.plugin.set_gpu_health(gpu.index, True)
