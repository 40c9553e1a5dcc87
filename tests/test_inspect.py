"""Inspect CLI: data model, annotation precedence, summary/details output."""

import io
import json

from gpushare_amd import consts
from gpushare_amd.cli import inspect as insp
from gpushare_amd.cluster.kubeclient import FakeKubeClient

from helpers import make_pod


def _cluster():
    kube = FakeKubeClient("node-a")
    kube.nodes["node-a"]["status"] = {
        "capacity": {consts.RESOURCE_COUNT: "2", consts.RESOURCE_NAME: "16"},
        "allocatable": {consts.RESOURCE_COUNT: "2", consts.RESOURCE_NAME: "16"},
        "addresses": [{"type": "InternalIP", "address": "10.0.0.5"}],
    }
    kube.nodes["plain-node"] = {
        "metadata": {"name": "plain-node", "labels": {}},
        "status": {"capacity": {}, "allocatable": {}},
    }
    return kube


def test_get_allocation_precedence():
    # allocation-map annotation wins
    pod = make_pod(
        "p", 6, gpu_idx=0,
        extra_annotations={
            consts.ANN_GPUSHARE_ALLOCATION: json.dumps({"c0": {"1": 4}, "c1": {"1": 2}})
        },
    )
    assert insp.get_allocation(pod) == {1: 6}
    # fallback to IDX
    assert insp.get_allocation(make_pod("p", 3, gpu_idx=1)) == {1: 3}
    # no annotation -> pending bucket
    pod = make_pod("p", 3)
    del pod["metadata"]["annotations"][consts.ENV_RESOURCE_INDEX]
    assert insp.get_allocation(pod) == {insp.PENDING: 3}


def test_summary_output():
    kube = _cluster()
    kube.add_pod(make_pod("a", 3, gpu_idx=0, assigned="true", phase="Running"))
    kube.add_pod(make_pod("b", 2, gpu_idx=0, assigned="true", phase="Running"))
    kube.add_pod(make_pod("c", 4, gpu_idx=1, assigned="true", phase="Running"))
    # terminal pod must not count
    kube.add_pod(make_pod("dead", 8, gpu_idx=1, phase="Failed"))
    out = io.StringIO()
    assert insp.main([], kube=kube, out=out) == 0
    text = out.getvalue()
    assert "GPU0(Allocated/Total)" in text
    assert "5/8" in text       # GPU0: 3+2 of 8
    assert "4/8" in text       # GPU1
    assert "9/16 (56%)" in text
    assert "plain-node" not in text   # not a sharing node


def test_details_output():
    kube = _cluster()
    kube.add_pod(make_pod("a", 3, gpu_idx=0, assigned="true", phase="Running"))
    pending = make_pod("pend", 2, phase="Pending")
    del pending["metadata"]["annotations"][consts.ENV_RESOURCE_INDEX]
    kube.add_pod(pending)
    out = io.StringIO()
    assert insp.main(["-d"], kube=kube, out=out) == 0
    text = out.getvalue()
    assert "NAME:       node-a" in text
    assert "IPADDRESS:  10.0.0.5" in text
    assert "Pending(Allocated)" in text
    assert "pend" in text
    assert "Allocated : 5 (31%)" in text


def test_node_filter_and_missing():
    kube = _cluster()
    out = io.StringIO()
    assert insp.main(["no-such-node"], kube=kube, out=out) == 1
    assert "No shared-GPU nodes" in out.getvalue()


def test_unit_inference_mi355x_gib():
    """287 GiB per GPU must NOT be misread as MiB (reference threshold 100
    would; ours is 1024)."""
    kube = FakeKubeClient("node-a")
    kube.nodes["node-a"]["status"] = {
        "capacity": {},
        "allocatable": {
            consts.RESOURCE_COUNT: "8",
            consts.RESOURCE_NAME: str(287 * 8),
        },
    }
    infos = insp.build_node_infos(kube)
    assert insp.infer_memory_unit(infos) == consts.GIB
    # true-MiB node
    kube.nodes["node-a"]["status"]["allocatable"][consts.RESOURCE_NAME] = str(
        287 * 1024 * 8
    )
    infos = insp.build_node_infos(kube)
    assert insp.infer_memory_unit(infos) == consts.MIB


def test_inspect_over_http_apiserver():
    from gpushare_amd.cluster.fakeapiserver import FakeApiServer

    kube = _cluster()
    kube.add_pod(make_pod("h", 4, gpu_idx=1, assigned="true", phase="Running"))
    api = FakeApiServer(store=kube).start()
    try:
        out = io.StringIO()
        assert insp.main(["--api-url", api.url], kube=None, out=out) == 0
        assert "4/8" in out.getvalue()
    finally:
        api.stop()


def test_gpushare_top_snapshot():
    """gpushare-top renders per-GPU and per-process rows from a source
    (mock source lacks usage APIs -> zero-usage snapshot still works)."""
    import io

    from gpushare_amd.cli import top
    from gpushare_amd.device.mock_source import MockSource

    class FakeUsageSource:
        def __init__(self):
            self._gpus = MockSource.from_spec("2x288GiB").devices()

        def devices(self):
            return self._gpus

        def vram_usage(self):
            return {0: 5 << 30, 1: 0}

        def process_usage(self):
            return {
                0: [{"pid": 4242, "vram_bytes": 5 << 30, "gtt_bytes": 0,
                     "cu_occupancy": 12, "container_name": "tenant-a",
                     "gfx_engine_ns": 10}],
                1: [],
            }

    out = io.StringIO()
    assert top.main([], source=FakeUsageSource(), out=out) == 0
    text = out.getvalue()
    assert "5.0GiB/288.0GiB" in text
    assert "4242" in text and "tenant-a" in text


def test_summary_heterogeneous_capacities():
    """With the topology annotation present, the summary shows each GPU's
    REAL capacity instead of total/count (reference behavior)."""
    import io
    import json as _json

    from gpushare_amd.cli import inspect as insp

    node = {
        "metadata": {
            "name": "hetero",
            "annotations": {
                consts.ANN_NODE_TOPOLOGY: _json.dumps(
                    {"unit": "GiB", "per_gpu_units": [288, 96],
                     "xgmi": [[1], [0]]}
                )
            },
        },
        "status": {
            "allocatable": {
                consts.RESOURCE_COUNT: "2",
                consts.RESOURCE_NAME: "384",
            },
            "addresses": [{"type": "InternalIP", "address": "10.0.0.9"}],
        },
    }
    pod = make_pod("p", mem=64, gpu_idx=0, node="hetero")
    pod["metadata"]["annotations"][consts.ENV_ASSIGNED_FLAG] = "true"
    info = insp.NodeInfo(node, [pod])
    out = io.StringIO()
    insp.display_summary([info], out=out)
    text = out.getvalue()
    assert "64/288" in text
    assert "0/96" in text


def test_gpushare_top_json_output():
    import io
    import json

    from gpushare_amd.cli import top
    from gpushare_amd.device.mock_source import MockSource

    class Src:
        def __init__(self):
            self._g = MockSource.from_spec("1x16GiB").devices()

        def devices(self):
            return self._g

        def vram_usage(self):
            return {0: 1 << 30}

        def process_usage(self):
            return {0: []}

    out = io.StringIO()
    assert top.main(["-o", "json"], source=Src(), out=out) == 0
    d = json.loads(out.getvalue())
    assert d["gpus"][0]["vram_used_bytes"] == 1 << 30


def test_inspect_retries_transient_apiserver_errors(capsys):
    """Reference parity (podinfo.go:64-70): a transient error on the
    first list must be retried, not surfaced to the kubectl user."""
    from gpushare_amd.cli import inspect as inspect_cli

    kube = FakeKubeClient("node-a")
    kube.patch_node_status(
        "node-a",
        {"status": {"allocatable": {consts.RESOURCE_NAME: "288"},
                    "capacity": {consts.RESOURCE_NAME: "288"}}},
    )
    fails = {"n": 2}
    real_list_nodes = kube.list_nodes

    def flaky_list_nodes():
        if fails["n"] > 0:
            fails["n"] -= 1
            raise ConnectionError("transient apiserver hiccup")
        return real_list_nodes()

    kube.list_nodes = flaky_list_nodes
    infos = inspect_cli.build_node_infos(kube)
    assert [i.name for i in infos] == ["node-a"]
    assert fails["n"] == 0


def test_inspect_retry_budget_exhausts():
    from gpushare_amd.cli import inspect as inspect_cli

    calls = {"n": 0}

    def always_fails():
        calls["n"] += 1
        raise ConnectionError("down")

    import pytest as _pytest

    with _pytest.raises(ConnectionError):
        inspect_cli._with_retries(always_fails, retries=5, interval=0.001)
    assert calls["n"] == 5


def test_json_output_mode():
    """-o json: machine-readable dump with per-device totals/pods and
    cluster aggregates."""
    kube = _cluster()
    kube.add_pod(make_pod("t1", 4, gpu_idx=0, assigned="true",
                          phase="Running"))
    kube.add_pod(make_pod("t2", 6, gpu_idx=1, assigned="true",
                          phase="Running"))
    out = io.StringIO()
    rc = insp.main([], kube=kube, out=out)
    assert rc == 0  # default table still works

    out = io.StringIO()
    rc = insp.main(["-o", "json"], kube=kube, out=out)
    assert rc == 0
    data = json.loads(out.getvalue())
    assert data["unit"] in ("GiB", "MiB")
    assert data["cluster"]["gpu_mem_total"] == 16
    assert data["cluster"]["gpu_mem_used"] == 10
    node = data["nodes"][0]
    assert node["name"] == "node-a"
    assert node["devices"]["0"]["used"] == 4
    assert node["devices"]["1"]["used"] == 6
    pods1 = node["devices"]["1"]["pods"]
    assert [(p["name"], p["gpu_mem"]) for p in pods1] == [("t2", 6)]
