#!/usr/bin/env python3
"""Inventory-scale experiment: ListAndWatch payloads from GiB to MiB grain.

SURVEY §7 hard part 2: the MI355X fake-device count is 9–18× the
reference's hardware (288 grains/GPU at GiB; 294,912 at MiB — 2.36M
devices on an 8-GPU node).  This measures, for each granularity:

  - native codec pre-encode time (devlist_codec.cpp) + payload size
  - health-flip re-encode time (one GPU's grains flipped)
  - full gRPC round-trip: plugin ListAndWatch stream -> stub-kubelet
    client parse, over a real unix socket
  - plugin RSS

Usage: python benchmarks/inventory_scale.py [--mib-gpus N]
Writes a JSON summary to stdout; runs CPU-only (mock source).
"""

from __future__ import annotations

import argparse
import json
import os
import resource
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gpushare_amd import consts
from gpushare_amd.allocator import Allocator
from gpushare_amd.cluster.kubeclient import FakeKubeClient
from gpushare_amd.cluster.podmanager import PodManager
from gpushare_amd.device.fakedev import FakeDeviceTable, make_codec
from gpushare_amd.device.mock_source import MockSource
from gpushare_amd.deviceplugin.server import GPUSharePlugin
from gpushare_amd.deviceplugin.stubkubelet import StubKubelet


def measure(spec: str, unit: str, stream: bool = True,
            client_max_mb: int = 4, numa: bool = False) -> dict:
    gpus = MockSource.from_spec(spec).devices()
    t0 = time.perf_counter()
    table = FakeDeviceTable.build(gpus, unit)
    t_table = time.perf_counter() - t0
    t0 = time.perf_counter()
    codec = make_codec(table.ids, numa=table.numa_of if numa else None)
    t_codec = time.perf_counter() - t0
    t0 = time.perf_counter()
    payload = codec.encode([])
    t_encode = time.perf_counter() - t0
    flip = table.gpu_fake_indices(0)
    t0 = time.perf_counter()
    payload_flip = codec.encode(list(flip))
    t_flip = time.perf_counter() - t0

    out = {
        "spec": spec,
        "unit": unit,
        "numa_topology": numa,
        "devices": len(table),
        "payload_mb": round(len(payload) / 1e6, 2),
        "table_build_ms": round(t_table * 1e3, 1),
        "codec_build_ms": round(t_codec * 1e3, 1),
        "encode_ms": round(t_encode * 1e3, 1),
        "health_flip_encode_ms": round(t_flip * 1e3, 1),
    }
    assert len(payload_flip) >= len(payload)

    if stream:
        kube = FakeKubeClient(node_name="scale-node")
        pm = PodManager(kube, "scale-node", kubelet_client=None,
                        query_kubelet=False, apiserver_retries=0)
        with tempfile.TemporaryDirectory(prefix="gpushare-scale-") as sockdir:
            plugin = GPUSharePlugin(
                gpus, Allocator(gpus, pm), unit=unit, socket_dir=sockdir,
                allow_oversize_inventory=True,
            )
            kubelet = StubKubelet(sockdir)
            kubelet.start()
            try:
                plugin.serve()
                client = kubelet.wait_for_plugin(
                    consts.RESOURCE_NAME, timeout=30,
                    max_receive_mb=client_max_mb,
                )
                t0 = time.perf_counter()
                devices = client.wait_for_devices(
                    min_count=len(table), timeout=120
                )
                out["stream_roundtrip_ms"] = round(
                    (time.perf_counter() - t0) * 1e3, 1
                )
                out["client_devices"] = len(devices)
            finally:
                plugin.stop()
                kubelet.stop()
    out["rss_mb"] = round(
        resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024, 1
    )
    return out


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--mib-gpus", type=int, default=8,
                   help="GPUs for the MiB-grain case (8 = full node)")
    args = p.parse_args()
    results = [
        measure("8x288GiB", consts.GIB),
        # modern Device.topology NUMA hints: quantify the payload cost
        measure("8x288GiB", consts.GIB, numa=True),
        # MiB grain exceeds the stock kubelet 4 MiB gRPC limit (the plugin
        # refuses it by default); measured here with a raised client limit
        measure("1x288GiB", consts.MIB, client_max_mb=64),
        measure(f"{args.mib_gpus}x288GiB", consts.MIB, client_max_mb=128),
    ]
    print(json.dumps(results, indent=1))
    return 0


if __name__ == "__main__":
    sys.exit(main())
