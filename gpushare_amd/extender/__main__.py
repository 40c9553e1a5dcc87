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
    n = 0
    for node in kube.list_nodes().get("items", []):
        alloc = node.get("status", {}).get("allocatable", {})
        count = int(alloc.get(consts.RESOURCE_COUNT, 0) or 0)
        total = int(alloc.get(consts.RESOURCE_NAME, 0) or 0)
        if count > 0 and total > 0:
            name = node["metadata"]["name"]
            extender.register_node(name, [total // count] * count)
            n += 1
    return n


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="gpushare-scheduler-extender")
    p.add_argument("--port", type=int, default=32766)
    p.add_argument("--resync-interval", type=float, default=30.0)
    p.add_argument("--api-url", default=None, help=argparse.SUPPRESS)
    args = p.parse_args(argv)
    logging.basicConfig(level=logging.INFO, stream=sys.stderr)

    kube = RestKubeClient(base_url=args.api_url) if args.api_url else RestKubeClient()
    extender = GPUShareExtender(kube, resync_interval=args.resync_interval)
    n = discover_nodes(kube, extender)
    extender.resync()
    log.info("serving binpack extender for %d shared-GPU node(s)", n)
    server = ExtenderServer(extender, port=args.port)
    server._thread.daemon = False
    server.start()
    print(f"READY {server.url}", flush=True)
    try:
        while True:
            time.sleep(args.resync_interval)
            discover_nodes(kube, extender)
            extender.resync()
    except KeyboardInterrupt:
        server.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
