"""Device-plugin daemon entrypoint (reference: cmd/nvidia/main.go).

Flag parity with the reference (main.go:15-26) minus the dead ``--mps``
flag (plumbed but never read anywhere in the reference — SURVEY §2 dead
code; deliberately not carried), plus MI355X-native additions:

  --mock-spec        run against fake GPUs (CI; BASELINE config 1)
  --deep-probe       active gfx950 canary health probe interval (seconds)
  --no-inject        don't add /dev/kfd+renderD DeviceSpecs (env-only mode)
  --no-informer      list-per-Allocate instead of the pod watch cache
  --memguard-dir     stage + inject the LD_PRELOAD VRAM budget enforcer
  --allow-oversize-inventory  serve >4MiB ListAndWatch (patched kubelets)

Env: NODE_NAME (required — reference crashes at package-import time if
unset, podmanager.go:52-55; we fail at startup with a clear message),
KUBECONFIG (optional; in-cluster config otherwise).
"""

from __future__ import annotations

import argparse
import logging
import os
import sys

from .. import consts
from ..cluster.kubeclient import KubeletClient, RestKubeClient
from ..device import create_source
from ..lifecycle import ManagerOptions, SharedGPUManager


def parse_args(argv=None):
    p = argparse.ArgumentParser(
        prog="amdgpushare-device-plugin",
        description="MI355X GPU-sharing device plugin (aliyun.com/gpu-mem)",
    )
    p.add_argument("--health-check", action="store_true",
                   help="watch amdsmi events (thermal/reset/ECC) for device health")
    p.add_argument("--deep-probe", type=float, default=0.0, metavar="SECONDS",
                   help="also run the gfx950 canary kernels (MFMA+VRAM) on "
                        "every GPU at this interval; 0 disables")
    p.add_argument("--probe-mode", default="subprocess",
                   choices=("subprocess", "inproc"),
                   help="run canary probes in a short-lived child process "
                        "(default; keeps the daemon's RSS at control-plane "
                        "size and survives a GPU-wedging probe) or in-process")
    p.add_argument("--memory-unit", default=consts.GIB,
                   choices=list(consts.VALID_MEMORY_UNITS),
                   help="granularity of the gpu-mem resource (reference "
                        "main.go:67-78 validation)")
    p.add_argument("--query-kubelet", action="store_true",
                   help="list pending pods from the kubelet read-only API "
                        "instead of the apiserver")
    p.add_argument("--kubelet-address", default="127.0.0.1")
    p.add_argument("--kubelet-port", type=int, default=10250)
    p.add_argument("--client-cert", default="")
    p.add_argument("--client-key", default="")
    p.add_argument("--token", default="")
    p.add_argument("--timeout", type=int, default=10,
                   help="kubelet client timeout (seconds)")
    p.add_argument("--socket-dir", default=consts.DEVICE_PLUGIN_PATH)
    p.add_argument("--cache-ttl", type=float, default=0.2,
                   help="pending-pod cache TTL (seconds); 0 = list per Allocate "
                        "(reference behavior)")
    p.add_argument("--mock-spec", default=None,
                   help="e.g. 1x8GiB: serve fake devices (no GPU needed)")
    p.add_argument("--no-inject", action="store_true",
                   help="skip /dev/kfd+/dev/dri DeviceSpec injection")
    p.add_argument("--no-informer", action="store_true",
                   help="disable the pod watch informer; every Allocate "
                        "lists pods remotely (reference behavior)")
    p.add_argument("--allow-oversize-inventory", action="store_true",
                   help="serve a ListAndWatch payload above the kubelet's "
                        "4 MiB gRPC receive default (requires a patched "
                        "kubelet; MiB grain on 288 GiB GPUs needs this)")
    p.add_argument("--memguard-dir", default="", metavar="HOSTDIR",
                   help="enable per-container VRAM budget enforcement: "
                        "copy libgpushare_memguard.so into this hostPath "
                        "dir and inject it (LD_PRELOAD) into every "
                        "allocated container; empty = advisory isolation "
                        "only (reference behavior without cGPU)")
    p.add_argument("--trace-file", default="", metavar="PATH",
                   help="append a JSONL record per Allocate (ts, units, "
                        "outcome, per-stage ms) for debugging")
    p.add_argument("--metrics-port", type=int, default=0,
                   help="serve Prometheus /metrics on this port (0 = off)")
    p.add_argument("--numa-topology", action="store_true",
                   help="advertise each grain's NUMA domain via the modern "
                        "Device.topology field (kubelet TopologyManager "
                        "NUMA alignment; requires k8s >= 1.17 semantics — "
                        "older kubelets skip the field)")
    p.add_argument("--selftest", action="store_true",
                   help="one full register→ListAndWatch→Allocate pass "
                        "against an in-process stub kubelet with mock "
                        "GPUs, then exit 0/1 (container HEALTHCHECK / CI "
                        "smoke; no cluster, no GPU needed)")
    p.add_argument("-v", "--verbose", action="count", default=0)
    return p.parse_args(argv)


def selftest() -> int:
    """Image smoke: the whole plumbing path on fake devices.

    Exercises exactly BASELINE config 1 (mock GPU + stub kubelet socket):
    gRPC registration, the pre-encoded ListAndWatch inventory, and one
    Allocate with env + device-node injection.  Returns 0 on success.
    """
    import tempfile

    from ..allocator import Allocator
    from ..cluster.kubeclient import FakeKubeClient
    from ..cluster.podmanager import PodManager
    from ..device.mock_source import MockSource
    from ..deviceplugin.server import GPUSharePlugin
    from ..deviceplugin.stubkubelet import StubKubelet

    log = logging.getLogger("selftest")
    with tempfile.TemporaryDirectory(prefix="gpushare-st-") as socket_dir:
        kube = FakeKubeClient(node_name="selftest-node")
        pm = PodManager(
            kube, "selftest-node", kubelet_client=kube.as_kubelet(),
            cache_ttl=0.0, kubelet_retries=0, kubelet_retry_interval=0.0,
            apiserver_retries=0, apiserver_retry_interval=0.0,
        )
        gpus = MockSource.from_spec("1x8GiB").devices()
        plugin = GPUSharePlugin(gpus, Allocator(gpus, pm),
                                socket_dir=socket_dir)
        kubelet = StubKubelet(socket_dir)
        kubelet.start()
        try:
            plugin.serve()
            client = kubelet.wait_for_plugin(consts.RESOURCE_NAME)
            devices = client.wait_for_devices(min_count=8)
            if len(devices) != 8:
                log.error("selftest: expected 8 fake devices, got %d",
                          len(devices))
                return 1
            # hand-rolled assumed pod (no test helpers in the image)
            import time as _time
            kube.add_pod({
                "metadata": {
                    "name": "selftest-pod", "namespace": "default",
                    "uid": "selftest-uid",
                    "annotations": {
                        consts.ENV_RESOURCE_INDEX: "0",
                        consts.ENV_RESOURCE_ASSUME_TIME: str(_time.time_ns()),
                        consts.ENV_ASSIGNED_FLAG: "false",
                    },
                },
                "spec": {
                    "nodeName": "selftest-node",
                    "containers": [{
                        "name": "c0",
                        "resources": {"limits": {consts.RESOURCE_NAME: "4"}},
                    }],
                },
                "status": {"phase": "Pending"},
            })
            ids = sorted(client.wait_for_devices(8))[:4]
            resp = client.allocate([ids])
            envs = resp.container_responses[0].envs
            if envs[consts.ENV_RESOURCE_INDEX] != "0":
                log.error("selftest: bad allocate envs: %s", dict(envs))
                return 1
            log.info("selftest OK: registered, streamed 8 devices, "
                     "allocated 4 units on GPU 0")
            return 0
        except Exception as e:  # noqa: BLE001
            log.error("selftest failed: %s", e)
            return 1
        finally:
            plugin.stop()
            kubelet.stop()


def main(argv=None) -> int:
    args = parse_args(argv)
    logging.basicConfig(
        level=logging.DEBUG if args.verbose else logging.INFO,
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
        stream=sys.stderr,
    )
    log = logging.getLogger("daemon")

    if args.selftest:
        return selftest()

    node_name = os.environ.get("NODE_NAME")
    if not node_name:
        log.error("NODE_NAME env is required (set via downward API in the "
                  "DaemonSet, deploy/device-plugin-ds.yaml)")
        return 2

    if args.metrics_port:
        from .. import metrics

        metrics.serve(args.metrics_port)

    memguard_path = ""
    if args.memguard_dir:
        # stage the packaged enforcer onto the hostPath the kubelet will
        # bind-mount into allocated containers
        import shutil

        import gpushare_amd

        src = os.path.join(
            os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
        )
        if not os.path.exists(src):
            log.error("--memguard-dir set but %s is not built", src)
            return 2
        os.makedirs(args.memguard_dir, exist_ok=True)
        memguard_path = os.path.join(
            args.memguard_dir, "libgpushare_memguard.so"
        )
        shutil.copy2(src, memguard_path)
        log.info("memguard staged at %s", memguard_path)

    source = create_source(args.mock_spec)
    kube = RestKubeClient()
    kubelet = None
    if args.query_kubelet:
        kubelet = KubeletClient(
            address=args.kubelet_address,
            port=args.kubelet_port,
            token=args.token or None,
            timeout=args.timeout,
            client_cert=args.client_cert,
            client_key=args.client_key,
        )

    mgr = SharedGPUManager(
        source,
        kube,
        node_name,
        kubelet_client=kubelet,
        options=ManagerOptions(
            memory_unit=args.memory_unit,
            query_kubelet=args.query_kubelet,
            health_check=args.health_check,
            deep_probe_interval=args.deep_probe,
            probe_mode=args.probe_mode,
            socket_dir=args.socket_dir,
            cache_ttl=args.cache_ttl,
            inject_devices=not args.no_inject,
            use_informer=not args.no_informer,
            memguard_path=memguard_path,
            allow_oversize_inventory=args.allow_oversize_inventory,
            trace_file=args.trace_file,
            numa_topology=args.numa_topology,
        ),
    )
    mgr.install_signal_handlers()
    log.info("starting gpushare device plugin on node %s", node_name)
    try:
        mgr.run()
    except RuntimeError as e:
        log.error("fatal: %s", e)
        return 1
    return 0


if __name__ == "__main__":
    sys.exit(main())
