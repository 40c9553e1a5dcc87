"""Full-scale 8×MI355X topology fixture through the real resolution stack.

VERDICT r1 item 2: the multi-GPU paths (KFD walk → amdsmi enumeration →
topology annotation → xGMI-aware extender split) had only been exercised on
2-GPU toys and a 1-GPU box.  These tests drive them with a realistic 8-OAM
node layout (tests/helpers_kfd.py): driver-hash gpu_ids, NON-sequential
render minors with a hole occupied by an unrelated DRM device, two NUMA
domains, 288 GiB HBM3E per GPU, full 7-link xGMI mesh — the fidelity bar
set by the reference's device resolution (nvidia.go:60-85), which our KFD
walker must exceed because ROCm minors really are non-sequential.
"""

from __future__ import annotations

import json
import os

import pytest

from gpushare_amd import consts
from gpushare_amd.cluster.kubeclient import FakeKubeClient
from gpushare_amd.cluster.podmanager import PodManager
from gpushare_amd.device import amdsmi_source, kfd_topology
from gpushare_amd.extender.binpack import NodeGPUState
from gpushare_amd.extender.core import GPUShareExtender

from helpers_kfd import (
    FakeSmi,
    GPU_IDS,
    MI355X_VRAM,
    RENDER_MINORS,
    UNIQUE_IDS,
    write_8gpu_topology,
)

NODE = "amd-node-8"


@pytest.fixture
def layout(tmp_path):
    return write_8gpu_topology(tmp_path)


@pytest.fixture
def source(layout, monkeypatch):
    """AmdSmiSource over the fixture tree + fake amdsmi shim (enumeration
    order deliberately different from KFD-node order)."""
    monkeypatch.setattr(
        amdsmi_source, "_load_shim", lambda: FakeSmi(layout)
    )
    return amdsmi_source.AmdSmiSource(
        topology_root=layout["topology_root"], drm_root=layout["drm_root"]
    )


# --------------------------------------------------------------------------- #
# KFD walker
# --------------------------------------------------------------------------- #
class TestResolve:
    def test_eight_gpus_resolved(self, layout):
        topo = kfd_topology.resolve(
            layout["topology_root"], layout["drm_root"]
        )
        assert sorted(topo) == sorted(GPU_IDS)
        for i, gpu_id in enumerate(GPU_IDS):
            t = topo[gpu_id]
            assert t.render_path == f"/dev/dri/renderD{RENDER_MINORS[i]}"
            # card index is offset by the unrelated card0 display device
            assert t.card_path == f"/dev/dri/card{i + 1}"
            assert t.vram_bytes == MI355X_VRAM
            assert t.numa_node == (0 if i < 4 else 1)
            assert t.gfx_target_version == 90500
            assert t.unique_id == UNIQUE_IDS[i]
            # full 7-link xGMI mesh
            assert sorted(t.xgmi_peer_gpu_ids) == sorted(
                g for g in GPU_IDS if g != gpu_id
            )

    def test_unrelated_drm_device_not_claimed(self, layout):
        """renderD133 (display adapter in the minor hole) must not be
        attributed to any GPU."""
        topo = kfd_topology.resolve(
            layout["topology_root"], layout["drm_root"]
        )
        claimed = {t.render_path for t in topo.values()}
        assert "/dev/dri/renderD133" not in claimed
        claimed_cards = {t.card_path for t in topo.values()}
        assert "/dev/dri/card0" not in claimed_cards

    def test_cpu_nodes_excluded(self, layout):
        nodes = kfd_topology.read_topology(layout["topology_root"])
        assert len(nodes) == 10
        assert sum(1 for n in nodes if n.is_gpu) == 8


# --------------------------------------------------------------------------- #
# amdsmi enumeration over the fixture (order != KFD-node order)
# --------------------------------------------------------------------------- #
class TestEnumerate:
    def test_physical_gpus(self, source, layout):
        gpus = source.devices()
        assert len(gpus) == 8
        order = FakeSmi(layout).order
        for plugin_idx, g in enumerate(gpus):
            src = order[plugin_idx]
            assert g.index == plugin_idx
            assert g.memory_bytes == MI355X_VRAM
            assert g.mem_units("GiB") == 288
            assert g.render_path == f"/dev/dri/renderD{RENDER_MINORS[src]}"
            assert g.card_path == f"/dev/dri/card{src + 1}"
            assert g.numa_node == (0 if src < 4 else 1)
            assert g.extras["rocr_uuid"] == f"GPU-{UNIQUE_IDS[src]:016x}"
            # full mesh in PLUGIN index space
            assert g.xgmi_peers == tuple(
                i for i in range(8) if i != plugin_idx
            )

    def test_fake_device_ids_fit_kubelet_limit(self, source):
        """8 × 288 GiB grains: every fake-device ID must stay within the
        63-char kubelet object-name limit."""
        from gpushare_amd.device.fakedev import FakeDeviceTable

        table = FakeDeviceTable.build(source.devices(), consts.GIB)
        assert len(table) == 8 * 288
        assert all(len(i) <= 63 for i in table.ids)
        assert len(set(table.ids)) == len(table.ids)


# --------------------------------------------------------------------------- #
# extender: xGMI-aware multi-GPU split on the full-scale node
# --------------------------------------------------------------------------- #
class TestExtenderOnFixture:
    def _register(self, source):
        """The production handshake: plugin publishes the topology
        annotation, extender discovers it."""
        gpus = source.devices()
        kube = FakeKubeClient(node_name=NODE)
        pm = PodManager(
            kube,
            NODE,
            kubelet_client=kube.as_kubelet(),
            cache_ttl=0.0,
            kubelet_retries=0,
            apiserver_retries=0,
        )
        pm.patch_topology(gpus)
        ann = kube.get_node(NODE)["metadata"]["annotations"][
            consts.ANN_NODE_TOPOLOGY
        ]
        topo = json.loads(ann)
        assert topo["per_gpu_units"] == [288] * 8
        ext = GPUShareExtender(kube, resync_interval=3600)
        ext.register_node(
            NODE, topo["per_gpu_units"], xgmi=[list(p) for p in topo["xgmi"]]
        )
        return kube, ext

    def test_span_two_gpus_xgmi_adjacent(self, source):
        kube, ext = self._register(source)
        state: NodeGPUState = ext.state.nodes[NODE]
        # fill every GPU to 200/288 so no single GPU holds 288
        for i in range(8):
            state.allocated[i] = 200
        split = state.best_fit_multi(176)
        assert split is not None
        assert len(split) == 2
        assert sum(split.values()) == 176
        # with a full mesh every pair is adjacent: 1 edge
        assert state._xgmi_edges(tuple(split)) == 1

    def test_32x72_binpack_first_fit(self, source):
        """BASELINE config 3: 32 pods × 72 GiB pack 4-per-GPU across the
        8 fixture GPUs at 100%."""
        kube, ext = self._register(source)
        state: NodeGPUState = ext.state.nodes[NODE]
        placements = []
        for _ in range(32):
            idx = state.best_fit(72)
            assert idx is not None
            state.allocated[idx] += 72
            placements.append(idx)
        assert state.total_allocated == 32 * 72
        per_gpu = [placements.count(i) for i in range(8)]
        assert per_gpu == [4] * 8
        assert state.best_fit(72) is None  # node exactly full

    def test_giant_pod_spans_minimal_adjacent_set(self, source):
        """A 1,000 GiB request (no 4-GPU subset shy of it) must take the
        minimal set (4 GPUs × 288 = 1,152) — all pairwise xGMI-linked."""
        kube, ext = self._register(source)
        state: NodeGPUState = ext.state.nodes[NODE]
        split = state.best_fit_multi(1000)
        assert split is not None
        assert len(split) == 4
        assert sum(split.values()) == 1000
        assert state._xgmi_edges(tuple(split)) == 6  # complete K4


# --------------------------------------------------------------------------- #
# REAL captured topology (tests/fixtures/kfd_snapshot_mi355x_1gpu)
# --------------------------------------------------------------------------- #
REAL_SNAPSHOT = os.path.join(
    os.path.dirname(__file__), "fixtures", "kfd_snapshot_mi355x_1gpu"
)


class TestRealSnapshot:
    """Ground truth from a gpurun MI355X box: 1 visible GPU out of 8,
    masked peers with EMPTY gpu_id files, render minor 144, 287.98 GiB."""

    def _resolve(self):
        return kfd_topology.resolve(
            os.path.join(REAL_SNAPSHOT, "kfd", "topology", "nodes"),
            os.path.join(REAL_SNAPSHOT, "drm"),
        )

    def test_exactly_one_visible_gpu(self):
        topo = self._resolve()
        assert list(topo) == [23660]

    def test_masked_peers_not_gpus(self):
        """Empty gpu_id files (the 7 masked OAMs) must parse as CPU-side
        nodes, never as 0-byte GPUs."""
        nodes = kfd_topology.read_topology(
            os.path.join(REAL_SNAPSHOT, "kfd", "topology", "nodes")
        )
        gpus = [n for n in nodes if n.is_gpu]
        assert len(gpus) == 1
        assert gpus[0].gpu_id == 23660

    def test_real_gpu_properties(self):
        t = self._resolve()[23660]
        assert t.render_path == "/dev/dri/renderD144"  # NOT 128+index
        assert t.gfx_target_version == 90500
        assert t.vram_bytes == 309_220_868_096  # 287.98 GiB, not clean 288
        # xGMI links point at MASKED nodes -> no resolvable peers
        assert t.xgmi_peer_gpu_ids == []

    def test_grain_count_287(self):
        """Real HBM size floors to 287 whole GiB grains (what round-1 GPU
        runs advertised)."""
        from gpushare_amd.device import PhysicalGPU
        from gpushare_amd.device.fakedev import FakeDeviceTable

        t = self._resolve()[23660]
        gpu = PhysicalGPU(
            index=0, uuid="amd-real", memory_bytes=t.vram_bytes
        )
        table = FakeDeviceTable.build([gpu], consts.GIB)
        assert len(table) == 287
        assert all(len(i) <= 63 for i in table.ids)

    def test_masked_mesh_visible_in_raw_links(self):
        """The full 7-link xGMI mesh is present in the raw io_links even
        when peers are masked — the basis for the 8-GPU synthetic fixture's
        shape."""
        nodes = {
            n.node_id: n
            for n in kfd_topology.read_topology(
                os.path.join(REAL_SNAPSHOT, "kfd", "topology", "nodes")
            )
        }
        gpu = nodes[4]
        assert sorted(gpu.xgmi_peer_nodes) == [2, 3, 5, 6, 7, 8, 9]
        assert gpu.pcie_peer_nodes == [0]


class TestNumaTopologyAdvertisement:
    def test_plugin_advertises_numa_per_grain(self, source):
        """--numa-topology: grains carry their GPU's NUMA domain through
        the Device.topology field (4 GPUs per socket on the fixture)."""
        from gpushare_amd.allocator import Allocator
        from gpushare_amd.cluster.podmanager import PodManager
        from gpushare_amd.deviceplugin import v1beta1 as api
        from gpushare_amd.deviceplugin.server import GPUSharePlugin

        gpus = source.devices()
        kube = FakeKubeClient(node_name=NODE)
        pm = PodManager(
            kube, NODE, kubelet_client=kube.as_kubelet(), cache_ttl=0.0,
            kubelet_retries=0, apiserver_retries=0,
        )
        plugin = GPUSharePlugin(
            gpus, Allocator(gpus, pm), socket_dir="/tmp",
            numa_topology=True,
        )
        resp = api.ListAndWatchResponse.FromString(
            plugin.encoded_device_list()
        )
        assert len(resp.devices) == 8 * 288
        per_gpu_numa = {}
        for d in resp.devices:
            gpu_idx = plugin.table.gpu_of[d.ID]
            numa_ids = [n.ID for n in d.topology.nodes]
            per_gpu_numa.setdefault(gpu_idx, set()).update(numa_ids)
        # every grain of one GPU advertises exactly its GPU's NUMA node
        for g in gpus:
            assert per_gpu_numa[g.index] == {g.numa_node}

    def test_numa_topology_off_by_default(self, source):
        from gpushare_amd.allocator import Allocator
        from gpushare_amd.cluster.podmanager import PodManager
        from gpushare_amd.deviceplugin import v1beta1 as api
        from gpushare_amd.deviceplugin.server import GPUSharePlugin

        gpus = source.devices()
        kube = FakeKubeClient(node_name=NODE)
        pm = PodManager(
            kube, NODE, kubelet_client=kube.as_kubelet(), cache_ttl=0.0,
            kubelet_retries=0, apiserver_retries=0,
        )
        plugin = GPUSharePlugin(
            gpus, Allocator(gpus, pm), socket_dir="/tmp"
        )
        resp = api.ListAndWatchResponse.FromString(
            plugin.encoded_device_list()
        )
        assert all(not d.HasField("topology") for d in resp.devices)


class TestNumaTiebreak:
    """On an 8-OAM full xGMI mesh every GPU pair ties on link count; the
    second placement key is host-memory locality (4 GPUs per socket)."""

    def test_split_prefers_same_socket(self, source):
        gpus = source.devices()
        state = NodeGPUState(
            NODE,
            [g.mem_units("GiB") for g in gpus],
            xgmi=[g.xgmi_peers for g in gpus],
            numa=[g.numa_node for g in gpus],
        )
        # equal free everywhere -> edges and free tie across all pairs;
        # numa must break the tie: the chosen pair shares a socket
        for i in range(8):
            state.allocated[i] = 200
        split = state.best_fit_multi(170)
        assert len(split) == 2
        domains = {state.numa[i] for i in split}
        assert len(domains) == 1, (
            f"split {sorted(split)} spans sockets {domains}"
        )

    def test_cross_socket_still_allowed_when_forced(self, source):
        gpus = source.devices()
        numa = [g.numa_node for g in gpus]
        state = NodeGPUState(
            NODE,
            [g.mem_units("GiB") for g in gpus],
            xgmi=[g.xgmi_peers for g in gpus],
            numa=numa,
        )
        # leave free memory on exactly one GPU per socket: a 2-GPU split
        # must go cross-socket rather than fail
        socket0 = next(i for i in range(8) if numa[i] == 0)
        socket1 = next(i for i in range(8) if numa[i] == 1)
        for i in range(8):
            state.allocated[i] = 288 if i not in (socket0, socket1) else 188
        split = state.best_fit_multi(180)
        assert split is not None
        assert sorted(split) == sorted([socket0, socket1])

    def test_unknown_numa_neutral(self):
        """All-unknown NUMA (old plugins, missing annotation) must not
        change pre-r2 behavior: tightest pack still wins."""
        state = NodeGPUState(
            "n", [16, 16, 16], allocated=[10, 8, 8],
            xgmi=[(1, 2), (0, 2), (0, 1)],
        )
        split = state.best_fit_multi(10)
        # no single fits? 16-10=6,8,8 free; 10 needs 2 GPUs; tightest
        # total picks the two 8-free GPUs (16 total) over 6+8=14? No:
        # smaller total wins -> {0,1} or {0,2} with 14. Edges all equal.
        assert len(split) == 2
        assert sum(split.values()) == 10
        assert 0 in split  # the 6-free GPU is in the tightest pair
