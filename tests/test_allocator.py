"""Allocator: match path, env contract, device injection, failure paths."""

from gpushare_amd import consts
from gpushare_amd.allocator import Allocator
from gpushare_amd.cluster.kubeclient import FakeKubeClient
from gpushare_amd.cluster.podmanager import PodManager
from gpushare_amd.device.mock_source import MockSource
from gpushare_amd.deviceplugin import v1beta1 as api

from helpers import make_pod


def _request(container_units: list[int], uuid="mock-00") -> "api.AllocateRequest":
    req = api.AllocateRequest()
    n = 0
    for units in container_units:
        cr = req.container_requests.add()
        for _ in range(units):
            cr.devicesIDs.append(f"{uuid}-_-{n}")
            n += 1
    return req


def _setup(spec="8x288GiB", node="node-a", **alloc_kw):
    kube = FakeKubeClient(node_name=node)
    pm = PodManager(
        kube,
        node,
        kubelet_client=kube.as_kubelet(),
        cache_ttl=0.0,
        kubelet_retries=0,
        kubelet_retry_interval=0.0,
        apiserver_retries=0,
        apiserver_retry_interval=0.0,
    )
    gpus = MockSource.from_spec(spec).devices()
    return kube, Allocator(gpus, pm, **alloc_kw)


def test_allocate_matches_assumed_pod():
    kube, alloc = _setup()
    kube.add_pod(make_pod("p1", 72, gpu_idx=3))
    resp = alloc.allocate(_request([72]))
    assert len(resp.container_responses) == 1
    envs = resp.container_responses[0].envs
    assert envs[consts.ENV_RESOURCE_INDEX] == "3"
    assert envs[consts.ENV_RESOURCE_BY_POD] == "72"
    assert envs[consts.ENV_RESOURCE_BY_CONTAINER] == "72"
    assert envs[consts.ENV_RESOURCE_BY_DEV] == "288"
    assert envs[consts.ENV_HIP_VISIBLE] == "0"
    assert envs[consts.ENV_ROCR_VISIBLE].startswith("GPU-")
    # pod marked assigned
    stored = kube.get_pod("default", "p1")
    assert stored["metadata"]["annotations"][consts.ENV_ASSIGNED_FLAG] == "true"


def test_allocate_injects_amd_device_nodes():
    kube, alloc = _setup()
    kube.add_pod(make_pod("p1", 16, gpu_idx=1))
    resp = alloc.allocate(_request([16]))
    paths = {d.host_path for d in resp.container_responses[0].devices}
    assert consts.DEV_KFD in paths
    assert "/dev/dri/renderD129" in paths  # mock GPU 1
    assert "/dev/dri/card1" in paths
    for d in resp.container_responses[0].devices:
        assert d.container_path == d.host_path
        assert d.permissions == "rw"


def test_allocate_multi_container_pod():
    kube, alloc = _setup()
    kube.add_pod(make_pod("p1", 6, gpu_idx=0, containers=[2, 4]))
    resp = alloc.allocate(_request([2, 4]))
    assert len(resp.container_responses) == 2
    assert resp.container_responses[0].envs[consts.ENV_RESOURCE_BY_CONTAINER] == "2"
    assert resp.container_responses[1].envs[consts.ENV_RESOURCE_BY_CONTAINER] == "4"
    for c in resp.container_responses:
        assert c.envs[consts.ENV_RESOURCE_BY_POD] == "6"


def test_allocate_fifo_oldest_pod_wins():
    kube, alloc = _setup()
    kube.add_pod(make_pod("newer", 8, gpu_idx=1, assume_time_ns=2000))
    kube.add_pod(make_pod("older", 8, gpu_idx=2, assume_time_ns=1000))
    resp = alloc.allocate(_request([8]))
    assert resp.container_responses[0].envs[consts.ENV_RESOURCE_INDEX] == "2"
    assert (
        kube.get_pod("default", "older")["metadata"]["annotations"][
            consts.ENV_ASSIGNED_FLAG
        ]
        == "true"
    )
    assert (
        kube.get_pod("default", "newer")["metadata"]["annotations"][
            consts.ENV_ASSIGNED_FLAG
        ]
        == "false"
    )


def test_allocate_no_match_returns_poisoned_env():
    kube, alloc = _setup()  # 8 GPUs -> no single-GPU fast path
    resp = alloc.allocate(_request([9]))
    envs = resp.container_responses[0].envs
    assert envs[consts.ENV_ROCR_VISIBLE] == "no-gpu-has-9GiB-to-run"
    assert envs[consts.ENV_HIP_VISIBLE] == "no-gpu-has-9GiB-to-run"
    assert envs[consts.ENV_RESOURCE_INDEX] == "-1"
    assert alloc.stats.snapshot()["failures"] == 1


def test_allocate_bad_annotation_poisons():
    kube, alloc = _setup()
    pod = make_pod("p1", 4, gpu_idx=0)
    pod["metadata"]["annotations"][consts.ENV_RESOURCE_INDEX] = "not-a-number"
    kube.add_pod(pod)
    resp = alloc.allocate(_request([4]))
    assert "no-gpu-has-4GiB-to-run" in resp.container_responses[0].envs[
        consts.ENV_ROCR_VISIBLE
    ]


def test_allocate_annotation_out_of_range_poisons():
    kube, alloc = _setup()
    kube.add_pod(make_pod("p1", 4, gpu_idx=42))
    resp = alloc.allocate(_request([4]))
    assert "no-gpu-has" in resp.container_responses[0].envs[consts.ENV_ROCR_VISIBLE]


def test_single_gpu_fast_path_no_pod_needed():
    kube, alloc = _setup(spec="1x288GiB")
    # no pods at all -> still succeeds on a single-GPU node (allocate.go:151-178)
    resp = alloc.allocate(_request([72]))
    envs = resp.container_responses[0].envs
    assert envs[consts.ENV_RESOURCE_INDEX] == "0"
    assert envs[consts.ENV_RESOURCE_BY_DEV] == "288"
    assert kube.patch_count == 0


def test_assumed_pod_priority_over_fast_path():
    """Reference ordering: the found-pod branch wins over single-GPU."""
    kube, alloc = _setup(spec="1x288GiB")
    kube.add_pod(make_pod("p1", 72, gpu_idx=0))
    alloc.allocate(_request([72]))
    assert kube.patch_count == 1  # went through the pod path


def test_patch_failure_poisons():
    kube, alloc = _setup()
    kube.add_pod(make_pod("p1", 4, gpu_idx=0))
    kube.fail_next_pod_patches = 10
    resp = alloc.allocate(_request([4]))
    assert "no-gpu-has" in resp.container_responses[0].envs[consts.ENV_ROCR_VISIBLE]


def test_disable_isolation_env():
    kube, alloc = _setup(disable_isolation=True)
    kube.add_pod(make_pod("p1", 4, gpu_idx=0))
    resp = alloc.allocate(_request([4]))
    assert resp.container_responses[0].envs[consts.ENV_CGPU_DISABLE] == "true"


def test_mib_unit():
    kube, alloc = _setup(unit=consts.MIB)
    kube.add_pod(make_pod("p1", 1024, gpu_idx=0))
    resp = alloc.allocate(_request([1024]))
    envs = resp.container_responses[0].envs
    assert envs[consts.ENV_RESOURCE_BY_DEV] == str(288 * 1024)


def test_stats_percentiles_populated():
    kube, alloc = _setup()
    for i in range(5):
        kube.add_pod(make_pod(f"p{i}", 4, gpu_idx=0))
        alloc.allocate(_request([4]))
    snap = alloc.stats.snapshot()
    assert snap["count"] == 5
    assert snap["failures"] == 0
    assert snap["p50_ms"] > 0


def test_memguard_injection():
    """--memguard-dir mode: Allocate mounts the enforcer read-only and sets
    LD_PRELOAD + the per-container byte budget; disable_isolation wins."""
    kube = FakeKubeClient(node_name="node-a")
    pm = PodManager(kube, "node-a", kubelet_client=kube.as_kubelet(),
                    cache_ttl=0.0, kubelet_retries=0, apiserver_retries=0)
    gpus = MockSource.from_spec("1x16GiB").devices()
    alloc = Allocator(gpus, pm, memguard_path="/var/lib/gpushare/libgpushare_memguard.so")
    resp = alloc.allocate(_request([4]))
    c = resp.container_responses[0]
    assert c.envs[consts.ENV_MEMGUARD_LIMIT] == str(4 << 30)
    assert c.envs["LD_PRELOAD"] == consts.MEMGUARD_CONTAINER_PATH
    mounts = [(m.host_path, m.container_path, m.read_only) for m in c.mounts]
    assert mounts == [("/var/lib/gpushare/libgpushare_memguard.so",
                       consts.MEMGUARD_CONTAINER_PATH, True)]

    alloc2 = Allocator(gpus, pm, memguard_path="/x.so", disable_isolation=True)
    c2 = alloc2.allocate(_request([4])).container_responses[0]
    assert consts.ENV_MEMGUARD_LIMIT not in c2.envs
    assert len(c2.mounts) == 0


def test_allocate_trace_file(tmp_path):
    import json as _json

    kube = FakeKubeClient(node_name="node-a")
    pm = PodManager(kube, "node-a", kubelet_client=kube.as_kubelet(),
                    cache_ttl=0.0, kubelet_retries=0, apiserver_retries=0)
    gpus = MockSource.from_spec("1x16GiB").devices()
    trace = tmp_path / "alloc.jsonl"
    alloc = Allocator(gpus, pm, trace_file=str(trace))
    alloc.allocate(_request([4]))
    alloc.allocate(_request([2]))
    lines = [_json.loads(l) for l in trace.read_text().splitlines()]
    assert len(lines) == 2
    assert lines[0]["ok"] is True and lines[0]["req_units"] == 4
    assert lines[0]["total_ms"] >= 0
