"""Multi-GPU xGMI-aware placement (MI355X-native; no reference counterpart —
the reference rejects any pod whose gpu-mem exceeds one GPU).

Covers: binpack multi-fit set selection (minimal set, xGMI preference,
tightest pack), extender assume/release with the allocation-map annotation,
the plugin's multi-device Allocate response, and the topology-annotation
publish/consume handshake.
"""

from __future__ import annotations

import json

from gpushare_amd import consts
from gpushare_amd.allocator import Allocator
from gpushare_amd.cluster import podutils
from gpushare_amd.cluster.kubeclient import FakeKubeClient
from gpushare_amd.cluster.podmanager import PodManager
from gpushare_amd.device.mock_source import MockSource
from gpushare_amd.deviceplugin import v1beta1 as api
from gpushare_amd.extender.binpack import BinpackState, NodeGPUState
from gpushare_amd.extender.core import GPUShareExtender

from helpers import make_pod

NODE = "node-a"
FULL_MESH_4 = [[1, 2, 3], [0, 2, 3], [0, 1, 3], [0, 1, 2]]


def _request(container_units, uuid="mock-00"):
    req = api.AllocateRequest()
    n = 0
    for units in container_units:
        cr = req.container_requests.add()
        for _ in range(units):
            cr.devicesIDs.append(f"{uuid}-_-{n}")
            n += 1
    return req


class TestBestFitMulti:
    def test_single_gpu_preferred(self):
        st = NodeGPUState(NODE, [16, 16], xgmi=[(1,), (0,)])
        assert st.best_fit_multi(10) == {0: 10}

    def test_split_when_no_single_fits(self):
        st = NodeGPUState(NODE, [16, 16], xgmi=[(1,), (0,)])
        split = st.best_fit_multi(24)
        assert split is not None
        assert sum(split.values()) == 24
        assert set(split) == {0, 1}

    def test_minimal_set_size(self):
        st = NodeGPUState(NODE, [16] * 4, xgmi=FULL_MESH_4)
        split = st.best_fit_multi(30)
        assert len(split) == 2
        split = st.best_fit_multi(34)
        assert len(split) == 3

    def test_xgmi_adjacency_preferred(self):
        # GPUs 0-1 linked, 2-3 linked, no cross links; 0 and 2 are
        # partially used so {0,2} would be the tightest pack — adjacency
        # must win over tightness
        st = NodeGPUState(
            NODE,
            [16, 16, 16, 16],
            allocated=[6, 0, 6, 0],
            xgmi=[(1,), (0,), (3,), (2,)],
        )
        split = st.best_fit_multi(20)
        assert set(split) in ({0, 1}, {2, 3})

    def test_tightest_pack_among_equal_adjacency(self):
        # full mesh: adjacency equal everywhere; prefer the pair with the
        # least total free (consume fragments first)
        st = NodeGPUState(
            NODE, [16] * 4, allocated=[10, 10, 8, 8], xgmi=FULL_MESH_4
        )
        split = st.best_fit_multi(12)
        assert split == {0: 6, 1: 6}  # tightest pair, least-free first

    def test_infeasible(self):
        st = NodeGPUState(NODE, [16, 16], xgmi=[(1,), (0,)])
        assert st.best_fit_multi(33) is None

    def test_filter_accepts_multi_fit(self):
        bs = BinpackState()
        bs.set_node(NODE, [16, 16], xgmi=[(1,), (0,)])
        assert bs.filter_nodes(24, [NODE]) == [NODE]
        assert bs.filter_nodes(33, [NODE]) == []

    def test_assume_release_multi(self):
        bs = BinpackState()
        bs.set_node(NODE, [16, 16], xgmi=[(1,), (0,)])
        split = bs.assume_multi(NODE, 24)
        assert sum(split.values()) == 24
        assert bs.packing()["allocated_units"] == 24
        bs.release_multi(NODE, split)
        assert bs.packing()["allocated_units"] == 0


class TestExtenderMultiGPU:
    def _extender(self, per_gpu=(16, 16), xgmi=((1,), (0,))):
        kube = FakeKubeClient(node_name=NODE)
        ext = GPUShareExtender(kube, resync_interval=3600)
        ext.register_node(NODE, list(per_gpu), xgmi=[list(p) for p in xgmi])
        return kube, ext

    def test_assume_writes_allocation_map(self):
        kube, ext = self._extender()
        pod = make_pod("big", node=NODE, mem=24)
        del pod["metadata"]["annotations"]  # fresh pod, extender writes them
        kube.add_pod(pod)
        idx = ext.assume(pod, NODE)
        assert idx == 0
        stored = kube.get_pod("default", "big")
        anns = stored["metadata"]["annotations"]
        assert anns[consts.ENV_RESOURCE_INDEX] == "0"
        alloc = json.loads(anns[consts.ANN_GPUSHARE_ALLOCATION])
        merged = {
            int(i): u for per in alloc.values() for i, u in per.items()
        }
        assert sum(merged.values()) == 24
        assert set(merged) == {0, 1}

    def test_assume_splits_map_across_containers(self):
        """the map must carve the split across GPU-requesting
        containers (spec order), not attribute everything to the first —
        per container the units sum to its own limit, per GPU to the
        split."""
        kube, ext = self._extender()
        pod = make_pod("big2c", node=NODE, mem=24, containers=[16, 8])
        del pod["metadata"]["annotations"]
        kube.add_pod(pod)
        assert ext.assume(pod, NODE) == 0
        anns = kube.get_pod("default", "big2c")["metadata"]["annotations"]
        alloc = json.loads(anns[consts.ANN_GPUSHARE_ALLOCATION])
        assert set(alloc) == {"c0", "c1"}
        assert sum(alloc["c0"].values()) == 16
        assert sum(alloc["c1"].values()) == 8
        per_gpu: dict = {}
        for per in alloc.values():
            for i, u in per.items():
                per_gpu[i] = per_gpu.get(i, 0) + u
        assert sorted(per_gpu.values()) == [8, 16]

    def test_single_gpu_pod_gets_no_map(self):
        kube, ext = self._extender()
        pod = make_pod("small", node=NODE, mem=8)
        del pod["metadata"]["annotations"]
        kube.add_pod(pod)
        assert ext.assume(pod, NODE) is not None
        anns = kube.get_pod("default", "small")["metadata"]["annotations"]
        assert consts.ANN_GPUSHARE_ALLOCATION not in anns

    def test_release_uses_map(self):
        kube, ext = self._extender()
        pod = make_pod("big", node=NODE, mem=24)
        del pod["metadata"]["annotations"]
        kube.add_pod(pod)
        ext.assume(pod, NODE)
        assert ext.state.packing()["allocated_units"] == 24
        ext.release(kube.get_pod("default", "big"), NODE)
        assert ext.state.packing()["allocated_units"] == 0

    def test_resync_rebuilds_multi_allocations(self):
        kube, ext = self._extender()
        pod = make_pod(
            "big",
            node=NODE,
            mem=24,
            extra_annotations={
                consts.ANN_GPUSHARE_ALLOCATION: json.dumps(
                    {"main": {"0": 16, "1": 8}}
                )
            },
        )
        pod["status"]["phase"] = "Running"
        kube.add_pod(pod)
        ext.resync()
        per_node = ext.state.packing()["per_node"][NODE]
        assert per_node == [16, 8]


class TestAllocatorMultiGPU:
    def test_multi_device_response(self):
        gpus = MockSource.from_spec("4x16GiB").devices()
        kube = FakeKubeClient(node_name=NODE)
        pm = PodManager(
            kube,
            NODE,
            kubelet_client=kube.as_kubelet(),
            cache_ttl=0.0,
            kubelet_retries=0,
            apiserver_retries=0,
        )
        alloc = Allocator(gpus, pm)
        kube.add_pod(
            make_pod(
                "big",
                node=NODE,
                mem=24,
                gpu_idx=1,
                extra_annotations={
                    consts.ANN_GPUSHARE_ALLOCATION: json.dumps(
                        {"main": {"1": 16, "2": 8}}
                    )
                },
            )
        )
        resp = alloc.allocate(_request([24]))
        c = resp.container_responses[0]
        assert c.envs[consts.ENV_RESOURCE_INDEX] == "1"
        assert c.envs[consts.ENV_HIP_VISIBLE] == "0,1"
        rocr = c.envs[consts.ENV_ROCR_VISIBLE].split(",")
        assert len(rocr) == 2
        dev_paths = [d.host_path for d in c.devices]
        assert consts.DEV_KFD in dev_paths
        renders = [p for p in dev_paths if "renderD" in p]
        assert len(renders) == 2  # one per bound GPU

    def test_multi_device_memguard_split_envs(self):
        """Per-GPU sub-budgets: every container gets a
        container-scoped token plus GPUSHARE_MEM_LIMIT_BYTES_PER_DEVICE
        caps that carve the extender's split across containers — a tenant
        can no longer concentrate the whole pod budget on one split
        member."""
        gpus = MockSource.from_spec("4x16GiB").devices()
        kube = FakeKubeClient(node_name=NODE)
        pm = PodManager(
            kube,
            NODE,
            kubelet_client=kube.as_kubelet(),
            cache_ttl=0.0,
            kubelet_retries=0,
            apiserver_retries=0,
        )
        alloc = Allocator(gpus, pm, memguard_path="/x/libmemguard.so")
        kube.add_pod(
            make_pod(
                "big2c",
                node=NODE,
                mem=24,
                gpu_idx=1,
                containers=[16, 8],
                extra_annotations={
                    consts.ANN_GPUSHARE_ALLOCATION: json.dumps(
                        {"c0": {"1": 16, "2": 8}}
                    )
                },
            )
        )
        resp = alloc.allocate(_request([16, 8]))
        c0, c1 = resp.container_responses
        gib = 1 << 30
        # container 0: 16 GiB total, all of it on visible ordinal 0 (GPU 1)
        assert c0.envs[consts.ENV_MEMGUARD_CONTAINER_TOKEN] == "c0"
        assert c0.envs[consts.ENV_MEMGUARD_LIMIT] == str(16 * gib)
        assert c0.envs[consts.ENV_MEMGUARD_PER_DEVICE] == f"{16 * gib},0"
        # container 1: 8 GiB total, all on visible ordinal 1 (GPU 2)
        assert c1.envs[consts.ENV_MEMGUARD_CONTAINER_TOKEN] == "c1"
        assert c1.envs[consts.ENV_MEMGUARD_LIMIT] == str(8 * gib)
        assert c1.envs[consts.ENV_MEMGUARD_PER_DEVICE] == f"0,{8 * gib}"
        # caps sum exactly to the split per GPU
        caps0 = [int(x) for x in c0.envs[consts.ENV_MEMGUARD_PER_DEVICE].split(",")]
        caps1 = [int(x) for x in c1.envs[consts.ENV_MEMGUARD_PER_DEVICE].split(",")]
        assert [a + b for a, b in zip(caps0, caps1)] == [16 * gib, 8 * gib]

    def test_single_gpu_memguard_has_no_per_device_env(self):
        gpus = MockSource.from_spec("2x8GiB").devices()
        kube = FakeKubeClient(node_name=NODE)
        pm = PodManager(
            kube,
            NODE,
            kubelet_client=kube.as_kubelet(),
            cache_ttl=0.0,
            kubelet_retries=0,
            apiserver_retries=0,
        )
        alloc = Allocator(gpus, pm, memguard_path="/x/libmemguard.so")
        kube.add_pod(make_pod("small", node=NODE, mem=4, gpu_idx=0))
        resp = alloc.allocate(_request([4]))
        c = resp.container_responses[0]
        assert consts.ENV_MEMGUARD_PER_DEVICE not in c.envs
        assert c.envs[consts.ENV_MEMGUARD_CONTAINER_TOKEN] == "c0"

    def test_gpu_split_helper(self):
        pod = make_pod(
            "p",
            mem=24,
            extra_annotations={
                consts.ANN_GPUSHARE_ALLOCATION: json.dumps(
                    {"a": {"0": 8}, "b": {"0": 8, "1": 8}}
                )
            },
        )
        assert podutils.gpu_split_from_pod(pod) == {0: 16, 1: 8}


class TestTopologyAnnotation:
    def test_publish_and_discover(self):
        gpus = MockSource.from_spec("4x16GiB").devices()
        kube = FakeKubeClient(node_name=NODE)
        pm = PodManager(
            kube, NODE, kubelet_client=None, query_kubelet=False
        )
        pm.patch_topology(gpus)
        node = kube.get_node(NODE)
        topo = json.loads(
            node["metadata"]["annotations"][consts.ANN_NODE_TOPOLOGY]
        )
        assert topo["per_gpu_units"] == [16, 16, 16, 16]
        assert topo["xgmi"][0] == [1, 2, 3]  # mock: full mesh

        # extender daemon's discover path consumes it
        kube.patch_node_status(
            NODE,
            {
                "status": {
                    "allocatable": {
                        consts.RESOURCE_COUNT: "4",
                        consts.RESOURCE_NAME: "64",
                    }
                }
            },
        )
        from gpushare_amd.extender.__main__ import discover_nodes

        ext = GPUShareExtender(kube, resync_interval=3600)
        assert discover_nodes(kube, ext) == 1
        st = ext.state.nodes[NODE]
        assert st.per_gpu_units == [16, 16, 16, 16]
        assert st.xgmi[0] == (1, 2, 3)
        # r2: per-GPU NUMA nodes ride the same annotation end to end
        assert "numa" in topo
        assert st.numa == [g.numa_node for g in gpus]
        # resync must not drop the numa list (r2 regression guard)
        ext.resync()
        assert ext.state.nodes[NODE].numa == st.numa


class TestAutoRelease:
    def test_deleted_pod_releases_reservation(self):
        """Production flow: bind → (pod deleted) → watch DELETED → release,
        no webhook release call; idempotent with an explicit release."""
        import time as _t

        from gpushare_amd.cluster.fakeapiserver import FakeApiServer
        from gpushare_amd.cluster.kubeclient import RestKubeClient
        from gpushare_amd.extender.core import make_auto_release_informer

        server = FakeApiServer(store=FakeKubeClient(node_name=NODE)).start()
        try:
            kube = RestKubeClient(base_url=server.url)
            ext = GPUShareExtender(kube, resync_interval=3600)
            ext.register_node(NODE, [16, 16], xgmi=[[1], [0]])
            inf = make_auto_release_informer(kube, ext).start()
            assert inf.wait_synced(5)

            pod = make_pod("doomed", node=NODE, mem=8)
            del pod["metadata"]["annotations"]
            server.store.add_pod(pod)
            assert ext.assume(server.store.get_pod("default", "doomed"), NODE) is not None
            assert ext.state.packing()["allocated_units"] == 8

            server.store.delete_pod("default", "doomed")
            deadline = _t.monotonic() + 5
            while _t.monotonic() < deadline:
                if ext.state.packing()["allocated_units"] == 0:
                    break
                _t.sleep(0.01)
            assert ext.state.packing()["allocated_units"] == 0

            # explicit release after the watch already released: no-op
            ext.release(pod, NODE)
            assert ext.state.packing()["allocated_units"] == 0
            inf.stop()
        finally:
            server.stop()


class TestPlacementRecordFallback:
    def test_release_stub_without_map_frees_full_split(self):
        """A release carrying only the primary-index annotation (e.g. a
        scheduler's delete stub) must still free the full multi-GPU split,
        via the extender's placement record from assume time."""
        kube = FakeKubeClient(node_name=NODE)
        ext = GPUShareExtender(kube, resync_interval=3600)
        ext.register_node(NODE, [16, 16], xgmi=[[1], [0]])
        pod = make_pod("big", node=NODE, mem=24)
        del pod["metadata"]["annotations"]
        kube.add_pod(pod)
        assert ext.assume(pod, NODE) == 0
        assert ext.state.packing()["allocated_units"] == 24

        stub = make_pod("big", node=NODE, mem=24, gpu_idx=0)
        stub["metadata"].pop("uid", None)  # caller stub, no uid / no map
        ext.release(stub, NODE)
        assert ext.state.packing()["allocated_units"] == 0
        assert ext.state.packing()["per_node"][NODE] == [0, 0]

    def test_resync_prunes_stale_placements(self):
        kube = FakeKubeClient(node_name=NODE)
        ext = GPUShareExtender(kube, resync_interval=3600)
        ext.register_node(NODE, [16, 16], xgmi=[[1], [0]])
        pod = make_pod("gone", node=NODE, mem=24)
        del pod["metadata"]["annotations"]
        kube.add_pod(pod)
        ext.assume(pod, NODE)
        kube.delete_pod("default", "gone")
        ext.resync()
        assert ext._placements == {}
        assert ext.state.packing()["allocated_units"] == 0


class TestHeterogeneousNode:
    def test_topology_drives_per_gpu_capacities(self):
        """[288,288,96] node: the extender must reject a 100 GiB pod from
        landing on the 96 GiB GPU and place it on a 288 GiB one — possible
        only because the plugin publishes REAL per-GPU capacities (the
        reference derives total/count and would misplace)."""
        gpus = MockSource.from_spec("288+288+96GiB").devices()
        kube = FakeKubeClient(node_name=NODE)
        pm = PodManager(kube, NODE, kubelet_client=None, query_kubelet=False)
        pm.patch_topology(gpus)
        kube.patch_node_status(
            NODE,
            {"status": {"allocatable": {
                consts.RESOURCE_COUNT: "3",
                consts.RESOURCE_NAME: str(288 + 288 + 96),
            }}},
        )
        from gpushare_amd.extender.__main__ import discover_nodes

        ext = GPUShareExtender(kube, resync_interval=3600)
        assert discover_nodes(kube, ext) == 1
        assert ext.state.nodes[NODE].per_gpu_units == [288, 288, 96]

        # fill both 288s so only the 96 has room; a 100 GiB pod must span
        for name, mem in (("a", 288), ("b", 288)):
            pod = make_pod(name, node=NODE, mem=mem)
            del pod["metadata"]["annotations"]
            kube.add_pod(pod)
            assert ext.assume(pod, NODE) is not None
        pod = make_pod("c", node=NODE, mem=100)
        del pod["metadata"]["annotations"]
        kube.add_pod(pod)
        assert ext.assume(pod, NODE) is None  # 96 < 100 and others full


class TestTopologyAnnotationFallback:
    def test_corrupt_annotation_falls_back_to_uniform(self):
        """A broken topology annotation must not break discovery — fall
        back to the reference's uniform total/count split."""
        kube = FakeKubeClient(node_name=NODE)
        kube.patch_node(NODE, {"metadata": {"annotations": {
            consts.ANN_NODE_TOPOLOGY: "{not json",
        }}})
        kube.patch_node_status(NODE, {"status": {"allocatable": {
            consts.RESOURCE_COUNT: "2", consts.RESOURCE_NAME: "32",
        }}})
        from gpushare_amd.extender.__main__ import discover_nodes

        ext = GPUShareExtender(kube, resync_interval=3600)
        assert discover_nodes(kube, ext) == 1
        assert ext.state.nodes[NODE].per_gpu_units == [16, 16]

    def test_wrong_length_annotation_ignored(self):
        kube = FakeKubeClient(node_name=NODE)
        kube.patch_node(NODE, {"metadata": {"annotations": {
            consts.ANN_NODE_TOPOLOGY: json.dumps(
                {"per_gpu_units": [16, 16, 16], "xgmi": []}
            ),
        }}})
        kube.patch_node_status(NODE, {"status": {"allocatable": {
            consts.RESOURCE_COUNT: "2", consts.RESOURCE_NAME: "32",
        }}})
        from gpushare_amd.extender.__main__ import discover_nodes

        ext = GPUShareExtender(kube, resync_interval=3600)
        assert discover_nodes(kube, ext) == 1
        assert ext.state.nodes[NODE].per_gpu_units == [16, 16]
