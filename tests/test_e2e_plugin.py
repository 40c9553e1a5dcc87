"""End-to-end over real unix-socket gRPC: mock GPUs + stub kubelet + plugin.

This is BASELINE config 1 ("mock-GPU mode on CPU: plugin registers 1 fake
8GiB device with a stub kubelet socket") plus health-flip and restart
coverage — the full register → ListAndWatch → Allocate plumbing with no
hardware and no cluster.
"""

import threading

import pytest

from gpushare_amd import consts
from gpushare_amd.allocator import Allocator
from gpushare_amd.cluster.kubeclient import FakeKubeClient
from gpushare_amd.cluster.podmanager import PodManager
from gpushare_amd.device.mock_source import MockSource
from gpushare_amd.deviceplugin.server import GPUSharePlugin
from gpushare_amd.deviceplugin.stubkubelet import StubKubelet

from helpers import make_pod


@pytest.fixture
def harness(tmp_socket_dir):
    """(kube, plugin, kubelet) wired together over tmp sockets."""
    built = {}

    def build(spec="1x8GiB"):
        kube = FakeKubeClient(node_name="node-a")
        pm = PodManager(
            kube,
            "node-a",
            kubelet_client=kube.as_kubelet(),
            cache_ttl=0.0,
            kubelet_retries=0,
            kubelet_retry_interval=0.0,
            apiserver_retries=0,
            apiserver_retry_interval=0.0,
        )
        gpus = MockSource.from_spec(spec).devices()
        plugin = GPUSharePlugin(
            gpus,
            Allocator(gpus, pm),
            socket_dir=tmp_socket_dir,
        )
        kubelet = StubKubelet(tmp_socket_dir)
        kubelet.start()
        plugin.serve()
        built.update(kube=kube, plugin=plugin, kubelet=kubelet)
        return kube, plugin, kubelet

    yield build
    if built:
        built["plugin"].stop()
        built["kubelet"].stop()


def test_config1_register_and_inventory(harness):
    kube, plugin, kubelet = harness("1x8GiB")
    client = kubelet.wait_for_plugin(consts.RESOURCE_NAME)
    devices = client.wait_for_devices(min_count=8)
    assert len(devices) == 8
    assert all(h == consts.HEALTHY for h in devices.values())
    assert all(i.startswith("mock-00-_-") for i in devices)


def test_config1_allocate_via_socket(harness):
    kube, plugin, kubelet = harness("1x8GiB")
    client = kubelet.wait_for_plugin()
    ids = sorted(client.wait_for_devices(8))[:4]
    kube.add_pod(make_pod("tenant-a", 4, gpu_idx=0))
    resp = client.allocate([ids])
    envs = resp.container_responses[0].envs
    assert envs[consts.ENV_RESOURCE_INDEX] == "0"
    assert envs[consts.ENV_RESOURCE_BY_POD] == "4"
    paths = {d.host_path for d in resp.container_responses[0].devices}
    assert consts.DEV_KFD in paths


def test_health_flip_marks_all_grains_and_recovers(harness):
    kube, plugin, kubelet = harness("2x8GiB")
    client = kubelet.wait_for_plugin()
    client.wait_for_devices(16)
    assert client.updates_received == 1

    plugin.set_gpu_health(1, healthy=False)
    devices = client.wait_for_update(min_updates=2)
    unhealthy = {i for i, h in devices.items() if h == consts.UNHEALTHY}
    # ALL 8 grains of GPU 1 flip (reference flips only one — SURVEY §3.3 bug)
    assert unhealthy == {f"mock-01-_-{j}" for j in range(8)}

    plugin.set_gpu_health(1, healthy=True)
    devices = client.wait_for_update(min_updates=3)
    assert all(h == consts.HEALTHY for h in devices.values())


def test_health_flip_all_gpus(harness):
    kube, plugin, kubelet = harness("2x8GiB")
    client = kubelet.wait_for_plugin()
    client.wait_for_devices(16)
    plugin.set_gpu_health(None, healthy=False)  # UUID-less event => all
    devices = client.wait_for_update(min_updates=2)
    assert all(h == consts.UNHEALTHY for h in devices.values())


def test_plugin_restart_reregisters(harness, tmp_socket_dir):
    kube, plugin, kubelet = harness("1x8GiB")
    client = kubelet.wait_for_plugin()
    client.wait_for_devices(8)
    # simulate the lifecycle manager's restart path (gpumanager.go:83-88)
    plugin.stop()
    plugin.start()
    plugin.register()
    client2 = kubelet.wait_for_plugin()
    assert client2.wait_for_devices(8)


def test_concurrent_allocates_distinct_pods(harness):
    """Two same-size pods pending: concurrent Allocates must claim distinct
    pods (SURVEY §7 hard part 1 — the disambiguation race)."""
    kube, plugin, kubelet = harness("8x8GiB")
    client = kubelet.wait_for_plugin()
    client.wait_for_devices(64)
    kube.add_pod(make_pod("a", 2, gpu_idx=1, assume_time_ns=1000))
    kube.add_pod(make_pod("b", 2, gpu_idx=5, assume_time_ns=2000))

    results = []

    def run():
        resp = client.allocate([["mock-00-_-0", "mock-00-_-1"]])
        results.append(resp.container_responses[0].envs[consts.ENV_RESOURCE_INDEX])

    t1, t2 = threading.Thread(target=run), threading.Thread(target=run)
    t1.start(); t2.start(); t1.join(); t2.join()
    # both pods end up assigned, on their own annotated GPUs
    assert sorted(results) == ["1", "5"]
    for name in ("a", "b"):
        assert (
            kube.get_pod("default", name)["metadata"]["annotations"][
                consts.ENV_ASSIGNED_FLAG
            ]
            == "true"
        )


def test_pre_start_container_rpc(tmp_socket_dir):
    """PreStartContainer: empty-response stub, options say not required
    (reference parity: server.go:191-193; GetDevicePluginOptions empty)."""
    from gpushare_amd import consts
    from gpushare_amd.allocator import Allocator
    from gpushare_amd.cluster.kubeclient import FakeKubeClient
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.device.mock_source import MockSource
    from gpushare_amd.deviceplugin import v1beta1 as api
    from gpushare_amd.deviceplugin.server import GPUSharePlugin
    from gpushare_amd.deviceplugin.stubkubelet import StubKubelet

    import grpc

    kube = FakeKubeClient("node-a")
    gpus = MockSource.from_spec("1x8GiB").devices()
    pm = PodManager(kube, "node-a", kubelet_client=kube.as_kubelet(),
                    cache_ttl=0.0)
    plugin = GPUSharePlugin(gpus, Allocator(gpus, pm),
                            socket_dir=tmp_socket_dir)
    kubelet = StubKubelet(tmp_socket_dir)
    kubelet.start()
    try:
        plugin.serve()
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME)
        assert client.options.pre_start_required is False
        with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
            pre_start = ch.unary_unary(
                api.METHOD_PRE_START,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=api.PreStartContainerResponse.FromString,
            )
            req = api.PreStartContainerRequest()
            req.devicesIDs.append(f"{gpus[0].uuid}-_-0")
            resp = pre_start(req, timeout=5)
            assert resp is not None
    finally:
        plugin.stop()
        kubelet.stop()


def test_get_preferred_allocation_binpacks_one_gpu(harness):
    """Modern kubelet RPC (k8s >= 1.19): preferred grains must come from
    ONE GPU — the tightest-fitting one — so kubelet grain bookkeeping
    matches the extender's whole-GPU packing."""
    import grpc

    from gpushare_amd.deviceplugin import v1beta1 as api

    kube, plugin, kubelet = harness("2x8GiB")
    client = kubelet.wait_for_plugin(consts.RESOURCE_NAME)
    devices = client.wait_for_devices(min_count=16)

    opts = client.options
    assert opts.get_preferred_allocation_available is True

    channel = grpc.insecure_channel(f"unix://{plugin.socket_path}")
    call = channel.unary_unary(
        api.METHOD_GET_PREFERRED,
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=api.PreferredAllocationResponse.FromString,
    )
    try:
        # GPU0 has 3 grains available, GPU1 has 8: a request for 3 must
        # take the tightest fit (all of GPU0's remainder)
        gpu0 = sorted(i for i in devices if plugin.table.gpu_of[i] == 0)
        gpu1 = sorted(i for i in devices if plugin.table.gpu_of[i] == 1)
        req = api.PreferredAllocationRequest()
        cr = req.container_requests.add()
        cr.available_deviceIDs.extend(gpu0[:3] + gpu1)
        cr.allocation_size = 3
        resp = call(req, timeout=5)
        chosen = list(resp.container_responses[0].deviceIDs)
        assert len(chosen) == 3
        assert {plugin.table.gpu_of[i] for i in chosen} == {0}

        # must_include is honored and the remainder stays on one GPU
        req2 = api.PreferredAllocationRequest()
        cr2 = req2.container_requests.add()
        cr2.available_deviceIDs.extend(gpu0 + gpu1)
        cr2.must_include_deviceIDs.append(gpu1[0])
        cr2.allocation_size = 4
        resp2 = call(req2, timeout=5)
        chosen2 = list(resp2.container_responses[0].deviceIDs)
        assert len(chosen2) == 4
        assert gpu1[0] in chosen2

        # oversize (no single GPU fits): spread, most-available first,
        # full count still returned
        req3 = api.PreferredAllocationRequest()
        cr3 = req3.container_requests.add()
        cr3.available_deviceIDs.extend(gpu0[:2] + gpu1[:3])
        cr3.allocation_size = 5
        resp3 = call(req3, timeout=5)
        assert len(resp3.container_responses[0].deviceIDs) == 5
    finally:
        channel.close()
