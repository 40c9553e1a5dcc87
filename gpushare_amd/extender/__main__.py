"""Standalone scheduler-extender daemon: `python -m gpushare_amd.extender`.

Discovers shared-GPU nodes from the apiserver (allocatable
``aliyun.com/gpu-count`` / ``gpu-mem``), rebuilds the ledger from pod
annotations, and serves the filter/bind webhook.
"""

from __future__ import annotations

import argparse
import logging
import sys
import time

from .. import consts
from ..cluster.kubeclient import RestKubeClient
from .core import GPUShareExtender
from .server import ExtenderServer

log = logging.getLogger("extender")


def discover_nodes(kube, extender) -> int:
    """Register shared-GPU nodes.  Prefer the plugin-published topology
    annotation (real per-GPU capacities + xGMI adjacency for multi-GPU
    placement); fall back to the uniform total/count split the reference
    assumes (nvidia.go:70-72)."""
    import json

    n = 0
    for node in kube.list_nodes().get("items", []):
        alloc = node.get("status", {}).get("allocatable", {})
        count = int(alloc.get(consts.RESOURCE_COUNT, 0) or 0)
        total = int(alloc.get(consts.RESOURCE_NAME, 0) or 0)
        if not (count > 0 and total > 0):
            continue
        name = node["metadata"]["name"]
        per_gpu, xgmi, numa = [total // count] * count, None, None
        raw = (node["metadata"].get("annotations") or {}).get(
            consts.ANN_NODE_TOPOLOGY
        )
        if raw:
            try:
                topo = json.loads(raw)
                if len(topo.get("per_gpu_units", [])) == count:
                    per_gpu = [int(u) for u in topo["per_gpu_units"]]
                    xgmi = topo.get("xgmi")
                    numa = topo.get("numa")
            except (ValueError, TypeError, KeyError) as e:
                log.warning("bad topology annotation on %s: %s", name, e)
        extender.register_node(name, per_gpu, xgmi=xgmi, numa=numa)
        n += 1
    return n


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="gpushare-scheduler-extender")
    p.add_argument("--port", type=int, default=32766)
    p.add_argument("--resync-interval", type=float, default=30.0)
    p.add_argument("--policy", choices=("binpack", "spread"),
                   default="binpack",
                   help="binpack: co-locate tightly (reference behavior); "
                        "spread: most-free GPU first — trades packing for "
                        "less interference (measured trade: "
                        "profiles/colocation_fairness_gpu_box.md)")
    p.add_argument("--no-watch", action="store_true",
                   help="disable the pod watch (auto-release of deleted "
                        "pods' reservations); rely on resync only")
    p.add_argument("--api-url", default=None, help=argparse.SUPPRESS)
    args = p.parse_args(argv)
    logging.basicConfig(level=logging.INFO, stream=sys.stderr)

    kube = RestKubeClient(base_url=args.api_url) if args.api_url else RestKubeClient()
    extender = GPUShareExtender(kube, resync_interval=args.resync_interval,
                                policy=args.policy)
    # initial discovery retries: the apiserver may lag the daemon at boot
    n = 0
    for attempt in range(30):
        try:
            n = discover_nodes(kube, extender)
            extender.resync()
            break
        except Exception as e:  # noqa: BLE001
            log.warning("initial node discovery failed (%s); retrying", e)
            time.sleep(min(2.0 * (attempt + 1), 10.0))

    informer = None
    if not args.no_watch:
        from .core import make_auto_release_informer

        informer = make_auto_release_informer(kube, extender).start()

    log.info("serving binpack extender for %d shared-GPU node(s)", n)
    server = ExtenderServer(extender, port=args.port).start()
    print(f"READY {server.url}", flush=True)
    try:
        while True:
            time.sleep(args.resync_interval)
            try:
                discover_nodes(kube, extender)
                extender.resync()
            except Exception as e:  # noqa: BLE001 — an apiserver blip must
                log.warning("resync failed: %s", e)   # not kill the daemon
    except KeyboardInterrupt:
        if informer is not None:
            informer.stop()
        server.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
