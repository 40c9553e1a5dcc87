"""Whole-node simulation on CPU: every component wired together.

daemon (SharedGPUManager + mock GPUs) ⇄ stub kubelet (gRPC)
extender webhook (HTTP) ⇄ fake apiserver (HTTP) ⇄ plugin's REST client

Flow mirrors deploy/demo/binpack-1.yaml: four 72 GiB pods land on one
288 GiB GPU; then the inspect CLI reads the same cluster state.
"""

import io
import threading

import pytest

from gpushare_amd import consts
from gpushare_amd.cli import inspect as insp
from gpushare_amd.cluster.fakeapiserver import FakeApiServer
from gpushare_amd.cluster.kubeclient import RestKubeClient
from gpushare_amd.device.mock_source import MockSource
from gpushare_amd.deviceplugin.stubkubelet import StubKubelet
from gpushare_amd.extender.core import GPUShareExtender
from gpushare_amd.extender.server import ExtenderClient, ExtenderServer
from gpushare_amd.lifecycle import ManagerOptions, SharedGPUManager

NODE = "sim-node"


@pytest.fixture
def stack(tmp_socket_dir):
    api = FakeApiServer().start()
    api.store.node_name = NODE
    api.store.nodes = {
        NODE: {
            "metadata": {"name": NODE, "labels": {}},
            "status": {"capacity": {}, "allocatable": {}},
        }
    }
    kube = RestKubeClient(base_url=api.url)

    source = MockSource.from_spec("2x288GiB")
    mgr = SharedGPUManager(
        source,
        RestKubeClient(base_url=api.url),
        NODE,
        options=ManagerOptions(socket_dir=tmp_socket_dir, cache_ttl=0.05,
                               health_check=True),
    )
    kubelet = StubKubelet(tmp_socket_dir)
    kubelet.start()
    mgr_thread = threading.Thread(target=mgr.run, daemon=True)
    mgr_thread.start()

    extender = GPUShareExtender(RestKubeClient(base_url=api.url),
                                resync_interval=3600)
    extender.register_node(NODE, [288, 288])
    ext_server = ExtenderServer(extender).start()
    ext = ExtenderClient(ext_server.url)

    yield api, kube, kubelet, ext, extender, source
    mgr.shutdown()
    mgr_thread.join(timeout=5)
    kubelet.stop()
    ext.close()
    ext_server.stop()
    api.stop()
    kube.close()


def _gpu_pod(name, mem):
    return {
        "metadata": {"name": name, "namespace": "default"},
        "spec": {
            "nodeName": NODE,
            "containers": [
                {"name": "main",
                 "resources": {"limits": {consts.RESOURCE_NAME: str(mem)}}}
            ],
        },
        "status": {"phase": "Pending"},
    }


def test_binpack_demo_full_stack(stack):
    api, kube, kubelet, ext, extender, _source = stack
    client = kubelet.wait_for_plugin(consts.RESOURCE_NAME, timeout=15)
    devices = client.wait_for_devices(min_count=576, timeout=15)
    assert len(devices) == 576

    # node resources patched by the daemon
    node = kube.get_node(NODE)
    assert node["status"]["capacity"][consts.RESOURCE_COUNT] == "2"

    # schedule 4 × 72 GiB like the demo manifest: all must share GPU 0
    grains = sorted(devices)
    for i in range(4):
        name = f"binpack-1-{i}"
        api.store.add_pod(_gpu_pod(name, 72))
        assert ext.filter(api.store.get_pod("default", name), [NODE]) == [NODE]
        assert ext.bind("default", name, NODE) == ""
        resp = client.allocate([grains[i * 72:(i + 1) * 72]])
        envs = resp.container_responses[0].envs
        assert envs[consts.ENV_RESOURCE_INDEX] == "0", f"pod {i} not binpacked"
        assert envs[consts.ENV_ROCR_VISIBLE].startswith("GPU-")
        paths = {d.host_path for d in resp.container_responses[0].devices}
        assert consts.DEV_KFD in paths and "/dev/dri/renderD128" in paths

    # fifth pod: GPU0 full -> lands on GPU1
    api.store.add_pod(_gpu_pod("overflow", 72))
    assert ext.bind("default", "overflow", NODE) == ""
    resp = client.allocate([grains[288:360]])
    assert resp.container_responses[0].envs[consts.ENV_RESOURCE_INDEX] == "1"

    # all five assigned durably
    for name in [f"binpack-1-{i}" for i in range(4)] + ["overflow"]:
        anns = api.store.get_pod("default", name)["metadata"]["annotations"]
        assert anns[consts.ENV_ASSIGNED_FLAG] == "true"

    # packing report agrees: 360 of 576 used
    packing = ext.packing()
    assert packing["allocated_units"] == 5 * 72
    assert packing["per_node"][NODE] == [288, 72]

    # inspect CLI over the same apiserver shows the co-location
    # (set phase Running so the viewer counts them as active)
    for name in [f"binpack-1-{i}" for i in range(4)] + ["overflow"]:
        api.store.pods[("default", name)]["status"]["phase"] = "Running"
        api.store._reencode(("default", name))
    # inspect reads gpu-mem allocatable from node status; the kubelet would
    # maintain it from ListAndWatch — simulate that part
    kube.patch_node_status(
        NODE, {"status": {"allocatable": {consts.RESOURCE_NAME: "576"}}}
    )
    out = io.StringIO()
    assert insp.main(["--api-url", api.url], kube=None, out=out) == 0
    text = out.getvalue()
    assert "288/288" in text     # GPU0 fully allocated
    assert "72/288" in text      # GPU1 one tenant
    assert "360/576 (62%)" in text


def test_health_flip_visible_to_kubelet_full_stack(stack):
    """A thermal event on GPU 1 reaches the kubelet as 288 Unhealthy grains
    through source → monitor → plugin → ListAndWatch, then recovers."""
    from gpushare_amd.device import HealthEvent

    api, kube, kubelet, ext, extender, source = stack
    client = kubelet.wait_for_plugin(consts.RESOURCE_NAME, timeout=15)
    devices = client.wait_for_devices(576, timeout=15)
    assert all(h == consts.HEALTHY for h in devices.values())

    source.inject_health_event(
        HealthEvent(gpu_index=1, healthy=False, kind="thermal_throttle")
    )
    devices = client.wait_for_update(min_updates=2, timeout=15)
    unhealthy = {i for i, h in devices.items() if h == consts.UNHEALTHY}
    assert unhealthy == {f"mock-01-_-{j}" for j in range(288)}

    source.inject_health_event(HealthEvent(gpu_index=1, healthy=True))
    devices = client.wait_for_update(min_updates=3, timeout=15)
    assert all(h == consts.HEALTHY for h in devices.values())


def test_multigpu_span_full_stack(stack):
    """A 400 GiB pod (> one 288 GiB GPU) through the whole stack: the
    extender must split it over both GPUs (allocation-map annotation), the
    plugin's real gRPC Allocate must inject BOTH render nodes, and the
    inspect CLI must show the per-GPU split."""
    api, kube, kubelet, ext, extender, _source = stack
    # register with xGMI adjacency as the daemon's topology annotation would
    extender.register_node(NODE, [288, 288], xgmi=[[1], [0]])
    client = kubelet.wait_for_plugin(consts.RESOURCE_NAME, timeout=15)
    devices = client.wait_for_devices(min_count=576, timeout=15)
    grains = sorted(devices)

    api.store.add_pod(_gpu_pod("span", 400))
    assert ext.filter(api.store.get_pod("default", "span"), [NODE]) == [NODE]
    assert ext.bind("default", "span", NODE) == ""

    anns = api.store.get_pod("default", "span")["metadata"]["annotations"]
    alloc = anns[consts.ANN_GPUSHARE_ALLOCATION]
    import json as _json

    merged = {
        int(i): u
        for per in _json.loads(alloc).values()
        for i, u in per.items()
    }
    assert sum(merged.values()) == 400 and set(merged) == {0, 1}

    resp = client.allocate([grains[:400]])
    c = resp.container_responses[0]
    assert c.envs[consts.ENV_HIP_VISIBLE] == "0,1"
    assert len(c.envs[consts.ENV_ROCR_VISIBLE].split(",")) == 2
    paths = {d.host_path for d in c.devices}
    assert {"/dev/dri/renderD128", "/dev/dri/renderD129"} <= paths

    # inspect shows the split across both GPUs
    api.store.pods[("default", "span")]["status"]["phase"] = "Running"
    api.store._reencode(("default", "span"))
    kube.patch_node_status(
        NODE, {"status": {"allocatable": {consts.RESOURCE_NAME: "576"}}}
    )
    out = io.StringIO()
    assert insp.main(["--api-url", api.url], kube=None, out=out) == 0
    text = out.getvalue()
    assert "400/576" in text
