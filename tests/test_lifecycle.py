"""Lifecycle manager, watchers, health monitor, coredump, daemon CLI."""

import os
import signal
import threading
import time

import pytest

from gpushare_amd import consts
from gpushare_amd.cluster.kubeclient import FakeKubeClient
from gpushare_amd.coredump import coredump, stack_trace_all
from gpushare_amd.device import HealthEvent
from gpushare_amd.device.mock_source import MockSource
from gpushare_amd.deviceplugin.stubkubelet import StubKubelet
from gpushare_amd.health import HealthMonitor
from gpushare_amd.lifecycle import ManagerOptions, SharedGPUManager
from gpushare_amd.watchers import FSWatcher, IN_CREATE


def test_fswatcher_detects_create(tmp_path):
    w = FSWatcher()
    w.add(str(tmp_path))
    assert w.poll(timeout=0.05) == []
    (tmp_path / "kubelet.sock").write_text("")
    deadline = time.monotonic() + 2
    seen = []
    while time.monotonic() < deadline and not seen:
        seen = [e for e in w.poll(timeout=0.2) if e[1] == "kubelet.sock"]
    assert seen and seen[0][2] & IN_CREATE
    w.close()


def test_coredump_writes_stacks(tmp_path):
    path = coredump(str(tmp_path))
    assert path and os.path.exists(path)
    content = open(path).read()
    assert "MainThread" in content
    assert "test_coredump_writes_stacks" in stack_trace_all()


def _manager(tmp_socket_dir, spec="1x8GiB", **opt_kw):
    kube = FakeKubeClient(node_name="node-a")
    source = MockSource.from_spec(spec)
    opts = ManagerOptions(
        socket_dir=tmp_socket_dir, cache_ttl=0.0, **opt_kw
    )
    mgr = SharedGPUManager(source, kube, "node-a", options=opts)
    return kube, source, mgr


def test_manager_serves_and_patches_gpu_count(tmp_socket_dir):
    kube, source, mgr = _manager(tmp_socket_dir)
    kubelet = StubKubelet(tmp_socket_dir)
    kubelet.start()
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    try:
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME)
        assert len(client.wait_for_devices(8)) == 8
        node = kube.get_node("node-a")
        assert node["status"]["capacity"][consts.RESOURCE_COUNT] == "1"
    finally:
        mgr.shutdown()
        t.join(timeout=5)
        kubelet.stop()
    assert not t.is_alive()


def test_manager_restarts_on_kubelet_sock_recreation(tmp_socket_dir):
    kube, source, mgr = _manager(tmp_socket_dir)
    kubelet = StubKubelet(tmp_socket_dir)
    kubelet.start()
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    try:
        kubelet.wait_for_plugin(consts.RESOURCE_NAME, timeout=10)
        first_plugin = mgr.plugin
        # kubelet restart: socket recreated -> plugin must re-register
        kubelet.stop()
        kubelet = StubKubelet(tmp_socket_dir)
        kubelet.start()   # recreates kubelet.sock -> inotify IN_CREATE
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME, timeout=10)
        assert client.wait_for_devices(8)
        assert mgr.plugin is not first_plugin
    finally:
        mgr.shutdown()
        t.join(timeout=5)
        kubelet.stop()


def test_manager_sighup_restart(tmp_socket_dir):
    kube, source, mgr = _manager(tmp_socket_dir)
    kubelet = StubKubelet(tmp_socket_dir)
    kubelet.start()
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    try:
        kubelet.wait_for_plugin(consts.RESOURCE_NAME)
        first_plugin = mgr.plugin
        mgr._signals.put(signal.SIGHUP)
        deadline = time.monotonic() + 10
        while mgr.plugin is first_plugin and time.monotonic() < deadline:
            time.sleep(0.05)
        assert mgr.plugin is not first_plugin
        kubelet.wait_for_plugin(consts.RESOURCE_NAME)
    finally:
        mgr.shutdown()
        t.join(timeout=5)
        kubelet.stop()


def test_manager_refuses_zero_gpus(tmp_socket_dir):
    kube = FakeKubeClient(node_name="node-a")

    class EmptySource:
        def devices(self):
            return []

        def watch_health(self, stop):
            return iter(())

        def close(self):
            pass

    mgr = SharedGPUManager(
        EmptySource(), kube, "node-a",
        options=ManagerOptions(socket_dir="/tmp/nonexistent-ok"),
    )
    with pytest.raises(RuntimeError, match="no GPUs"):
        mgr.run(max_iterations=1)


def test_health_monitor_passive_flow(tmp_socket_dir):
    """MockSource-injected event reaches the plugin's health state."""
    from gpushare_amd.allocator import Allocator
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.deviceplugin.server import GPUSharePlugin

    kube = FakeKubeClient("node-a")
    source = MockSource.from_spec("2x8GiB")
    gpus = source.devices()
    pm = PodManager(kube, "node-a", kubelet_client=kube.as_kubelet(), cache_ttl=0)
    plugin = GPUSharePlugin(gpus, Allocator(gpus, pm), socket_dir=tmp_socket_dir)
    mon = HealthMonitor(source, plugin)
    mon.start()
    try:
        source.inject_health_event(
            HealthEvent(gpu_index=1, healthy=False, kind="thermal_throttle")
        )
        deadline = time.monotonic() + 5
        while not plugin._unhealthy_gpus and time.monotonic() < deadline:
            time.sleep(0.02)
        assert plugin._unhealthy_gpus == {1}
        source.inject_health_event(HealthEvent(gpu_index=1, healthy=True))
        while plugin._unhealthy_gpus and time.monotonic() < deadline:
            time.sleep(0.02)
        assert plugin._unhealthy_gpus == set()
    finally:
        mon.stop()


def test_daemon_cli_requires_node_name(monkeypatch):
    from gpushare_amd.cli.daemon import main

    monkeypatch.delenv("NODE_NAME", raising=False)
    assert main(["--mock-spec", "1x8GiB"]) == 2


def test_daemon_cli_flag_validation():
    from gpushare_amd.cli.daemon import parse_args

    args = parse_args(["--memory-unit", "MiB", "--query-kubelet"])
    assert args.memory_unit == "MiB"
    assert args.query_kubelet
    with pytest.raises(SystemExit):
        parse_args(["--memory-unit", "TiB"])


def test_allocation_state_survives_plugin_restart(tmp_socket_dir):
    """Crash-only invariant (reference §3.5): a pod assumed BEFORE a plugin
    restart allocates correctly AFTER it — all state lives in annotations,
    and a pod assigned before the restart is not re-matched after."""
    from helpers import make_pod

    kube, source, mgr = _manager(tmp_socket_dir, spec="2x16GiB")
    kubelet = StubKubelet(tmp_socket_dir)
    kubelet.start()
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    try:
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME)
        ids = sorted(client.wait_for_devices(32))

        # pod A allocated pre-restart
        kube.add_pod(make_pod("pre-a", 4, gpu_idx=0, node="node-a"))
        resp = client.allocate([ids[:4]])
        assert resp.container_responses[0].envs[consts.ENV_RESOURCE_INDEX] == "0"
        # pod B assumed pre-restart, allocated post-restart
        kube.add_pod(make_pod("pre-b", 6, gpu_idx=1, node="node-a"))

        mgr._signals.put(signal.SIGHUP)
        first_plugin = mgr.plugin
        deadline = time.monotonic() + 10
        while mgr.plugin is first_plugin and time.monotonic() < deadline:
            time.sleep(0.05)
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME, timeout=10)

        resp = client.allocate([ids[:6]])
        envs = resp.container_responses[0].envs
        assert envs[consts.ENV_RESOURCE_INDEX] == "1"  # pre-b, not pre-a
        anns = kube.get_pod("default", "pre-b")["metadata"]["annotations"]
        assert anns[consts.ENV_ASSIGNED_FLAG] == "true"
        # pre-a stays assigned exactly once (not re-matched)
        assert (
            kube.get_pod("default", "pre-a")["metadata"]["annotations"][
                consts.ENV_ASSIGNED_FLAG
            ]
            == "true"
        )
    finally:
        mgr.shutdown()
        t.join(timeout=5)
        kubelet.stop()


def test_health_monitor_subprocess_probe_flip_and_recover(
    tmp_socket_dir, monkeypatch
):
    """Deep probe in subprocess mode (r2 default: the daemon never maps
    the HIP runtime): a failing probe flips the GPU Unhealthy, a passing
    one recovers it — without gpushare_amd._canary ever being imported
    into this process."""
    import sys

    from gpushare_amd import health as health_mod
    from gpushare_amd.allocator import Allocator
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.deviceplugin.server import GPUSharePlugin

    kube = FakeKubeClient("node-a")
    source = MockSource.from_spec("2x8GiB")
    gpus = source.devices()
    pm = PodManager(
        kube, "node-a", kubelet_client=kube.as_kubelet(), cache_ttl=0
    )
    plugin = GPUSharePlugin(
        gpus, Allocator(gpus, pm), socket_dir=tmp_socket_dir
    )

    state = {"fail": True}

    def fake_probe(gpu_index, vram_probe_mb=32, timeout=60.0):
        if state["fail"] and gpu_index == 1:
            raise RuntimeError("canary subprocess rc=1: hipErrorNoDevice")
        return {"ok": True, "mfma_ok": True}

    monkeypatch.setattr(health_mod, "probe_in_subprocess", fake_probe)
    mon = HealthMonitor(
        source, plugin, deep_probe_interval=0.02, probe_mode="subprocess"
    )
    mon.start()
    try:
        deadline = time.monotonic() + 5
        while plugin._unhealthy_gpus != {1} and time.monotonic() < deadline:
            time.sleep(0.01)
        assert plugin._unhealthy_gpus == {1}

        state["fail"] = False
        while plugin._unhealthy_gpus and time.monotonic() < deadline:
            time.sleep(0.01)
        assert plugin._unhealthy_gpus == set()
        # the probe never pulled the HIP runtime into THIS process
        assert "gpushare_amd._canary" not in sys.modules
    finally:
        mon.stop()


def test_probe_in_subprocess_fails_loudly_without_gpu():
    """On a GPU-less host the child must exit nonzero and surface its
    stderr — never a silent healthy verdict."""
    import pytest as _pytest

    from gpushare_amd import health as health_mod

    try:
        import gpushare_amd._canary  # noqa: F401
    except ImportError:
        _pytest.skip("_canary extension not built")
    import torch

    if torch.cuda.is_available():
        _pytest.skip("GPU present; covered by the gpu-marked test")
    with _pytest.raises(RuntimeError, match="canary subprocess"):
        health_mod.probe_in_subprocess(0, vram_probe_mb=1)


def test_health_monitor_probe_timeout_flips_unhealthy(
    tmp_socket_dir, monkeypatch
):
    """A probe child that WEDGES (TimeoutExpired) must flip the GPU
    Unhealthy — a hung GPU is exactly when the canary must speak."""
    import subprocess as sp

    from gpushare_amd import health as health_mod
    from gpushare_amd.allocator import Allocator
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.deviceplugin.server import GPUSharePlugin

    kube = FakeKubeClient("node-a")
    source = MockSource.from_spec("1x8GiB")
    gpus = source.devices()
    pm = PodManager(
        kube, "node-a", kubelet_client=kube.as_kubelet(), cache_ttl=0
    )
    plugin = GPUSharePlugin(
        gpus, Allocator(gpus, pm), socket_dir=tmp_socket_dir
    )

    def wedged_probe(gpu_index, vram_probe_mb=32, timeout=60.0):
        raise sp.TimeoutExpired(cmd="canary", timeout=timeout)

    monkeypatch.setattr(health_mod, "probe_in_subprocess", wedged_probe)
    mon = HealthMonitor(
        source, plugin, deep_probe_interval=0.02, probe_mode="subprocess"
    )
    mon.start()
    try:
        deadline = time.monotonic() + 5
        while plugin._unhealthy_gpus != {0} and time.monotonic() < deadline:
            time.sleep(0.01)
        assert plugin._unhealthy_gpus == {0}
    finally:
        mon.stop()
