"""Real-MI355X tests (pytest -m gpu, run via gpurun).

These import the native extensions directly — a missing extension is a loud
failure on a GPU box, never a silent fallback.
"""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def smi():
    import gpushare_amd._amdsmi as smi

    assert smi.available(), "libamd_smi.so must dlopen on a GPU box"
    smi.init()
    yield smi
    # no shutdown: other module-scoped fixtures may still use it


@pytest.fixture(scope="module")
def source(smi):
    from gpushare_amd.device.amdsmi_source import AmdSmiSource

    return AmdSmiSource()


def test_amdsmi_enumerates_mi355x(smi):
    n = smi.device_count()
    assert n >= 1
    info = smi.device_info(0)
    # MI355X: 288 GiB HBM3E
    assert info["vram_total_bytes"] > 250 << 30, info
    assert info.get("uuid") or info.get("asic_serial"), info


def test_kfd_topology_real():
    from gpushare_amd.device import kfd_topology

    topo = kfd_topology.resolve()
    assert topo, "KFD topology must expose at least one GPU"
    for gpu_id, t in topo.items():
        assert t.render_path, f"GPU {gpu_id} has no render node"
        assert os.path.exists(t.render_path), t.render_path
        assert t.vram_bytes > 250 << 30
        assert t.gfx_target_version == 90500  # gfx950


def test_source_resolves_render_and_uuid(source):
    gpus = source.devices()
    assert gpus
    g = gpus[0]
    assert g.memory_bytes > 250 << 30
    assert g.render_path and os.path.exists(g.render_path)
    assert g.mem_units("GiB") >= 250
    assert len(g.uuid) <= 52
    rocr = g.extras.get("rocr_uuid")
    assert rocr is None or rocr.startswith("GPU-")


def test_amdsmi_count_matches_torch(smi):
    import torch

    assert torch.cuda.is_available()
    assert smi.device_count() == torch.cuda.device_count()


def test_canary_probe_mfma_and_vram():
    import gpushare_amd._canary as canary

    assert canary.device_count() >= 1
    result = canary.probe(0, vram_probe_mb=128, bandwidth=True)
    assert result["mfma_ok"], f"MFMA canary mismatch: {result}"
    assert result["vram_ok"], f"VRAM pattern mismatch: {result}"
    assert "gfx950" in result["arch"], result["arch"]
    # HBM3E streaming copy should comfortably exceed 1 TB/s
    assert result["hbm_copy_gbps"] > 1000, result


def test_ecc_counters_readable(smi):
    try:
        corr, uncorr = smi.ecc_count(0)
    except RuntimeError as e:
        pytest.skip(f"ecc counters unsupported on this box: {e}")
    assert uncorr == 0, f"GPU reports uncorrectable ECC errors: {uncorr}"


def test_e2e_allocate_and_tenant_workloads(source, tmp_path):
    """Full plugin round on real hardware, then two *concurrent* tenant
    processes sharing GPU 0 under the allocation's envs — the actual
    gpushare co-location semantics."""
    from gpushare_amd import consts
    from gpushare_amd.allocator import Allocator
    from gpushare_amd.cluster.kubeclient import FakeKubeClient
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.deviceplugin.server import GPUSharePlugin
    from gpushare_amd.deviceplugin.stubkubelet import StubKubelet
    from helpers import make_pod

    gpus = source.devices()
    sockdir = str(tmp_path / "dp")
    os.makedirs(sockdir)
    kube = FakeKubeClient(node_name="gpu-node")
    pm = PodManager(kube, "gpu-node", kubelet_client=kube.as_kubelet(), cache_ttl=0.0)
    plugin = GPUSharePlugin(gpus, Allocator(gpus, pm), socket_dir=sockdir)
    kubelet = StubKubelet(sockdir)
    kubelet.start()
    try:
        plugin.serve()
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME)
        total = sum(g.mem_units(consts.GIB) for g in gpus)
        devices = client.wait_for_devices(min_count=total)
        assert len(devices) == total  # 288 grains per MI355X

        # two 72 GiB tenants on GPU 0 (BASELINE config 2 shape)
        env_sets = []
        all_ids = sorted(devices)
        for i, name in enumerate(("tenant-a", "tenant-b")):
            kube.add_pod(make_pod(name, 72, gpu_idx=0, node="gpu-node"))
            ids = all_ids[i * 72 : (i + 1) * 72]
            resp = client.allocate([ids])
            envs = dict(resp.container_responses[0].envs)
            assert envs[consts.ENV_RESOURCE_INDEX] == "0"
            paths = {d.host_path for d in resp.container_responses[0].devices}
            assert consts.DEV_KFD in paths
            env_sets.append(envs)

        # run both tenant workloads concurrently against the shared GPU
        script = (
            "import torch; assert torch.cuda.is_available();"
            "x=torch.randn(1024,1024,device='cuda:0',dtype=torch.bfloat16);"
            "w=torch.randn(1024,1024,device='cuda:0',dtype=torch.bfloat16);"
            "y=(x@w).float().sum(); torch.cuda.synchronize();"
            "assert torch.isfinite(y); print('tenant ok', flush=True)"
        )
        procs = []
        for envs in env_sets:
            env = dict(os.environ)
            # injection-based isolation: all nodes already visible here, so
            # apply the env-based narrowing the response carries
            env[consts.ENV_ROCR_VISIBLE] = envs[consts.ENV_ROCR_VISIBLE]
            env[consts.ENV_HIP_VISIBLE] = envs[consts.ENV_HIP_VISIBLE]
            procs.append(
                subprocess.Popen(
                    [sys.executable, "-c", script],
                    env=env,
                    stdout=subprocess.PIPE,
                    stderr=subprocess.STDOUT,
                    cwd=REPO,
                )
            )
        for p in procs:
            out, _ = p.communicate(timeout=240)
            assert p.returncode == 0, out.decode()
            assert b"tenant ok" in out
    finally:
        plugin.stop()
        kubelet.stop()


def test_allocate_envs_drive_memguard_end_to_end(source, tmp_path):
    """The FULL enforcement chain on a real MI355X: a pod's Allocate
    response (not hand-crafted envs) configures the tenant — LD_PRELOAD
    mount + budget + pod/container scoping all come from the plugin, and
    the budget must then actually bind a PyTorch tenant."""
    from gpushare_amd import consts
    from gpushare_amd.allocator import Allocator
    from gpushare_amd.cluster.kubeclient import FakeKubeClient
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.deviceplugin.server import GPUSharePlugin
    from gpushare_amd.deviceplugin.stubkubelet import StubKubelet
    from helpers import make_pod

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    assert os.path.exists(lib), "memguard not built"

    gpus = source.devices()
    sockdir = str(tmp_path / "dp")
    os.makedirs(sockdir)
    kube = FakeKubeClient(node_name="gpu-node")
    pm = PodManager(
        kube, "gpu-node", kubelet_client=kube.as_kubelet(), cache_ttl=0.0
    )
    plugin = GPUSharePlugin(
        gpus, Allocator(gpus, pm, memguard_path=lib), socket_dir=sockdir
    )
    kubelet = StubKubelet(sockdir)
    kubelet.start()
    shm = None
    try:
        plugin.serve()
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME)
        all_ids = sorted(client.wait_for_devices(min_count=30))

        kube.add_pod(make_pod("budget-pod", 30, gpu_idx=0, node="gpu-node"))
        resp = client.allocate([all_ids[:30]])
        c = resp.container_responses[0]
        envs = dict(c.envs)
        # the response's enforcement contract
        assert envs[consts.ENV_MEMGUARD_LIMIT] == str(30 << 30)
        assert envs["LD_PRELOAD"] == consts.MEMGUARD_CONTAINER_PATH
        mounts = [(m.host_path, m.container_path) for m in c.mounts]
        assert (lib, consts.MEMGUARD_CONTAINER_PATH) in mounts
        pod_uid = envs[consts.ENV_MEMGUARD_POD_UID]
        token = envs[consts.ENV_MEMGUARD_CONTAINER_TOKEN]
        shm = f"/dev/shm/gpushare.memguard.{pod_uid}.{token}"

        script = (
            "import torch;"
            "free, total = torch.cuda.mem_get_info();"
            "assert total <= 30<<30, f'not clamped: {total}';"
            "a = torch.empty(8<<30, dtype=torch.uint8, device='cuda:0');"
            "exc = None\n"
            "try:\n"
            "    b = torch.empty(31<<30, dtype=torch.uint8, device='cuda:0')\n"
            "except torch.OutOfMemoryError:\n"
            "    print('E2E_MEMGUARD_OK', flush=True)\n"
            "else:\n"
            "    print('E2E_MEMGUARD_FAIL', flush=True)\n"
        )
        env = dict(os.environ)
        for key in (
            consts.ENV_MEMGUARD_LIMIT,
            consts.ENV_MEMGUARD_POD_UID,
            consts.ENV_MEMGUARD_CONTAINER_TOKEN,
            consts.ENV_ROCR_VISIBLE,
            consts.ENV_HIP_VISIBLE,
        ):
            if key in envs:
                env[key] = envs[key]
        # in k8s the kubelet bind-mounts host_path at container_path; here
        # both are this host, so preload the mount's SOURCE directly
        env["LD_PRELOAD"] = lib
        out = subprocess.run(
            [sys.executable, "-c", script],
            env=env,
            capture_output=True,
            text=True,
            timeout=240,
            cwd=REPO,
        )
        assert "E2E_MEMGUARD_OK" in out.stdout, (
            f"stdout={out.stdout!r} stderr={out.stderr[-1500:]!r}"
        )
    finally:
        if shm and os.path.exists(shm):
            os.unlink(shm)
        plugin.stop()
        kubelet.stop()


def test_manager_e2e_on_gpu(source, tmp_path):
    """Full lifecycle manager with the real device source on an MI355X."""
    import threading

    from gpushare_amd import consts
    from gpushare_amd.cluster.kubeclient import FakeKubeClient
    from gpushare_amd.deviceplugin.stubkubelet import StubKubelet
    from gpushare_amd.lifecycle import ManagerOptions, SharedGPUManager

    sockdir = str(tmp_path / "dp")
    kube = FakeKubeClient(node_name="gpu-node")
    mgr = SharedGPUManager(
        source, kube, "gpu-node",
        options=ManagerOptions(socket_dir=sockdir, cache_ttl=0.0,
                               health_check=True),
    )
    kubelet = StubKubelet(sockdir)
    os.makedirs(sockdir, exist_ok=True)
    kubelet.start()
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    try:
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME, timeout=30)
        n = len(client.wait_for_devices(min_count=1, timeout=30))
        total = sum(g.mem_units("GiB") for g in source.devices())
        assert n == total
        node = kube.get_node("gpu-node")
        assert node["status"]["capacity"][consts.RESOURCE_COUNT] == str(
            len(source.devices())
        )
    finally:
        mgr.shutdown()
        t.join(timeout=10)
        kubelet.stop()


def test_health_monitor_deep_probe_on_gpu(source, tmp_path):
    """Active canary probing against the real GPU: healthy stays healthy."""
    import time as _time

    from gpushare_amd.allocator import Allocator
    from gpushare_amd.cluster.kubeclient import FakeKubeClient
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.deviceplugin.server import GPUSharePlugin
    from gpushare_amd.health import HealthMonitor

    gpus = source.devices()
    kube = FakeKubeClient("gpu-node")
    pm = PodManager(kube, "gpu-node", kubelet_client=kube.as_kubelet(),
                    cache_ttl=0.0)
    plugin = GPUSharePlugin(gpus, Allocator(gpus, pm),
                            socket_dir=str(tmp_path))
    mon = HealthMonitor(source, plugin, deep_probe_interval=0.5,
                        probe_vram_mb=16)
    mon.start()
    try:
        _time.sleep(2.5)  # several probe cycles
        assert plugin._unhealthy_gpus == set()
        assert mon._probe_failed == set()
    finally:
        mon.stop()


@pytest.mark.gpu
def test_topology_annotation_from_real_device(source):
    """patch_topology publishes the real per-GPU capacity + xGMI list from
    KFD; the extender's discover path must accept it round-trip."""
    import json

    from gpushare_amd import consts
    from gpushare_amd.cluster.kubeclient import FakeKubeClient
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.extender.__main__ import discover_nodes
    from gpushare_amd.extender.core import GPUShareExtender

    gpus = source.devices()
    kube = FakeKubeClient(node_name="gpu-node")
    pm = PodManager(kube, "gpu-node", kubelet_client=None, query_kubelet=False)
    pm.patch_topology(gpus)
    topo = json.loads(
        kube.get_node("gpu-node")["metadata"]["annotations"][
            consts.ANN_NODE_TOPOLOGY
        ]
    )
    assert topo["per_gpu_units"][0] >= 200  # MI355X: ~288 GiB advertised
    assert len(topo["xgmi"]) == len(gpus)

    kube.patch_node_status(
        "gpu-node",
        {"status": {"allocatable": {
            consts.RESOURCE_COUNT: str(len(gpus)),
            consts.RESOURCE_NAME: str(sum(topo["per_gpu_units"])),
        }}},
    )
    ext = GPUShareExtender(kube, resync_interval=3600)
    assert discover_nodes(kube, ext) == 1
    assert ext.state.nodes["gpu-node"].per_gpu_units == topo["per_gpu_units"]


@pytest.mark.gpu
def test_memguard_enforces_vram_budget():
    """LD_PRELOAD enforcement on a real MI355X: a 2 GiB budget lets a
    1 GiB tensor through, clamps hipMemGetInfo to the budget, and turns
    an over-budget allocation into an ordinary torch OOM."""
    import os
    import subprocess
    import sys

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    assert os.path.exists(lib), "memguard not built"
    script = r"""
import torch
assert torch.cuda.is_available()
free, total = torch.cuda.mem_get_info()
assert total <= (2 << 30), f"hipMemGetInfo not clamped: total={total}"
a = torch.empty(1 << 30, dtype=torch.uint8, device="cuda:0")  # 1 GiB: fits
try:
    b = torch.empty(2 << 30, dtype=torch.uint8, device="cuda:0")
except torch.OutOfMemoryError:
    print("MEMGUARD_OK")
else:
    print("MEMGUARD_FAIL no OOM")
"""
    env = dict(os.environ)
    env["LD_PRELOAD"] = lib
    env["GPUSHARE_MEM_LIMIT_BYTES"] = str(2 << 30)
    out = subprocess.run(
        [sys.executable, "-c", script],
        env=env,
        capture_output=True,
        text=True,
        timeout=180,
    )
    assert "MEMGUARD_OK" in out.stdout, (
        f"stdout={out.stdout!r} stderr={out.stderr[-2000:]!r}"
    )


@pytest.mark.gpu
def test_memguard_budget_shared_across_processes():
    """Two processes under ONE pod budget (shm counter): while A holds
    1.5 GiB of a 2 GiB budget, B's 1 GiB allocation OOMs; after A exits
    (budget repaid), B's retry succeeds."""
    import os
    import subprocess
    import sys
    import time
    import uuid as uuid_mod

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    pod_uid = f"test-{uuid_mod.uuid4().hex[:12]}"
    shm_path = f"/dev/shm/gpushare.memguard.{pod_uid}.c"
    env = dict(os.environ)
    env["LD_PRELOAD"] = lib
    env["GPUSHARE_MEM_LIMIT_BYTES"] = str(2 << 30)
    env["GPUSHARE_POD_UID"] = pod_uid

    holder = subprocess.Popen(
        [sys.executable, "-c", (
            "import torch, time, sys;"
            "a=torch.empty(3<<29, dtype=torch.uint8, device='cuda:0');"
            "print('HELD', flush=True); time.sleep(60)"
        )],
        env=env, stdout=subprocess.PIPE, text=True,
    )
    try:
        line = holder.stdout.readline()
        assert "HELD" in line, f"holder failed: {line}"
        probe = (
            "import torch;\n"
            "try:\n"
            "    b=torch.empty(1<<30, dtype=torch.uint8, device='cuda:0')\n"
            "    print('ALLOC_OK', flush=True)\n"
            "except torch.OutOfMemoryError:\n"
            "    print('ALLOC_OOM', flush=True)\n"
        )
        out = subprocess.run(
            [sys.executable, "-c", probe], env=env,
            capture_output=True, text=True, timeout=180,
        )
        assert "ALLOC_OOM" in out.stdout, (
            f"expected shared-budget OOM: {out.stdout!r} {out.stderr[-800:]!r}"
        )
    finally:
        holder.terminate()
        holder.wait(timeout=30)
    # holder's exit repaid its reservation -> the same alloc now fits
    deadline = time.monotonic() + 30
    ok = False
    while time.monotonic() < deadline:
        out = subprocess.run(
            [sys.executable, "-c", probe], env=env,
            capture_output=True, text=True, timeout=180,
        )
        if "ALLOC_OK" in out.stdout:
            ok = True
            break
        time.sleep(2)
    if os.path.exists(shm_path):
        os.unlink(shm_path)
    assert ok, "budget was not repaid after holder exit"


@pytest.mark.gpu
def test_eight_tenants_shared_gpu_with_memguard():
    """The end goal of the framework: 8 co-located tenants on ONE MI355X,
    each enforced to a 30 GiB budget, all running real GEMMs concurrently;
    every tenant succeeds inside its budget and sees the clamped VRAM."""
    import os
    import subprocess
    import sys
    import uuid as uuid_mod

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    script = (
        "import os, torch;"
        "free, total = torch.cuda.mem_get_info();"
        "assert total <= 30<<30, f'not clamped: {total}';"
        "x = torch.randn(8192, 8192, device='cuda:0', dtype=torch.bfloat16);"
        "w = torch.randn(8192, 8192, device='cuda:0', dtype=torch.bfloat16);"
        "y = (x @ w).float().sum(); torch.cuda.synchronize();"
        "assert torch.isfinite(y);"
        "print('TENANT_OK', os.environ['T_ID'], flush=True)"
    )
    procs = []
    shm_paths = []
    for i in range(8):
        env = dict(os.environ)
        pod_uid = f"tenant{i}-{uuid_mod.uuid4().hex[:8]}"
        env["LD_PRELOAD"] = lib
        env["GPUSHARE_MEM_LIMIT_BYTES"] = str(30 << 30)  # 8×30 < 288 GiB
        env["GPUSHARE_POD_UID"] = pod_uid
        env["T_ID"] = str(i)
        shm_paths.append(f"/dev/shm/gpushare.memguard.{pod_uid}.c")
        procs.append(
            subprocess.Popen(
                [sys.executable, "-c", script],
                env=env, stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT, text=True,
            )
        )
    failures = []
    for i, p in enumerate(procs):
        out, _ = p.communicate(timeout=300)
        if p.returncode != 0 or "TENANT_OK" not in out:
            failures.append((i, p.returncode, out[-1500:]))
    for sp in shm_paths:
        if os.path.exists(sp):
            os.unlink(sp)
    assert not failures, failures


@pytest.mark.gpu
def test_memguard_expandable_segments_vmm_budget():
    """PyTorch's expandable_segments allocator is built on the HIP VMM
    family (hipMemCreate/hipMemMap) — the round-1 escape hatch.  Under a
    4 GiB budget: a 1 GiB tensor fits, a further 6 GiB tensor must OOM
    through the interposed hipMemCreate, and usage must be repaid on
    free."""
    import os
    import subprocess
    import sys
    import uuid as uuid_mod

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    assert os.path.exists(lib), "memguard not built"
    script = r"""
import torch
assert torch.cuda.is_available()
a = torch.empty(1 << 30, dtype=torch.uint8, device="cuda:0")
try:
    b = torch.empty(6 << 30, dtype=torch.uint8, device="cuda:0")
except torch.OutOfMemoryError:
    pass
else:
    raise SystemExit("VMM_FAIL: 6 GiB fit inside a 4 GiB budget")
del a
torch.cuda.empty_cache()
# repaid: a 3 GiB tensor fits again
c = torch.empty(3 << 30, dtype=torch.uint8, device="cuda:0")
c.fill_(7)
torch.cuda.synchronize()
print("VMM_MEMGUARD_OK")
"""
    pod_uid = f"vmm-{uuid_mod.uuid4().hex[:8]}"
    env = dict(os.environ)
    env["LD_PRELOAD"] = lib
    env["GPUSHARE_MEM_LIMIT_BYTES"] = str(4 << 30)
    env["GPUSHARE_POD_UID"] = pod_uid
    env["PYTORCH_HIP_ALLOC_CONF"] = "expandable_segments:True"
    out = subprocess.run(
        [sys.executable, "-c", script],
        env=env,
        capture_output=True,
        text=True,
        timeout=180,
    )
    shm = f"/dev/shm/gpushare.memguard.{pod_uid}.c"
    if os.path.exists(shm):
        os.unlink(shm)
    assert "VMM_MEMGUARD_OK" in out.stdout, (
        f"stdout={out.stdout!r} stderr={out.stderr[-2000:]!r}"
    )


@pytest.mark.gpu
def test_memguard_per_device_cap_enforced():
    """The per-ordinal sub-budget (multi-GPU split enforcement) must OOM
    an allocation that fits the container total but exceeds the device's
    cap.  On a 1-GPU box: total 6 GiB, ordinal-0 cap 2 GiB — a 3 GiB
    tensor must fail, a 1 GiB tensor must fit."""
    import os
    import subprocess
    import sys
    import uuid as uuid_mod

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    assert os.path.exists(lib), "memguard not built"
    script = r"""
import torch
free, total = torch.cuda.mem_get_info()
assert total <= (2 << 30), f"per-device clamp missing: total={total}"
a = torch.empty(1 << 30, dtype=torch.uint8, device="cuda:0")
try:
    b = torch.empty(3 << 30, dtype=torch.uint8, device="cuda:0")
except torch.OutOfMemoryError:
    print("PER_DEVICE_CAP_OK")
else:
    print("PER_DEVICE_CAP_FAIL")
"""
    pod_uid = f"devcap-{uuid_mod.uuid4().hex[:8]}"
    env = dict(os.environ)
    env["LD_PRELOAD"] = lib
    env["GPUSHARE_MEM_LIMIT_BYTES"] = str(6 << 30)
    env["GPUSHARE_MEM_LIMIT_BYTES_PER_DEVICE"] = str(2 << 30)
    env["GPUSHARE_POD_UID"] = pod_uid
    out = subprocess.run(
        [sys.executable, "-c", script],
        env=env,
        capture_output=True,
        text=True,
        timeout=180,
    )
    shm = f"/dev/shm/gpushare.memguard.{pod_uid}.c"
    if os.path.exists(shm):
        os.unlink(shm)
    assert "PER_DEVICE_CAP_OK" in out.stdout, (
        f"stdout={out.stdout!r} stderr={out.stderr[-2000:]!r}"
    )


@pytest.mark.gpu
def test_process_list_attribution(source):
    """amdsmi process list: a live torch process on GPU 0 must appear with
    its VRAM footprint (the raw data for pod-level usage attribution).
    amdsmi reports HOST-namespace pids, so from inside a container the
    entry is matched by its VRAM signature, not pid."""
    import subprocess
    import sys
    import time

    marker = (1 << 28) + (7 << 20)   # distinctive 263 MB footprint
    script = (
        f"import torch, time;"
        f"x = torch.empty({marker}, dtype=torch.uint8, device='cuda:0');"
        f"print('UP', flush=True); time.sleep(40)"
    )
    proc = subprocess.Popen(
        [sys.executable, "-c", script], stdout=subprocess.PIPE, text=True
    )
    try:
        assert "UP" in proc.stdout.readline()
        deadline = time.monotonic() + 25
        found = None
        while time.monotonic() < deadline and found is None:
            for p in source.process_usage().get(0, []):
                if p["vram_bytes"] >= marker:
                    found = p
                    break
            time.sleep(1)
        assert found is not None, (
            f"no process with >= {marker} B VRAM in amdsmi list: "
            f"{source.process_usage().get(0)}"
        )
    finally:
        proc.terminate()
        proc.wait(timeout=30)


@pytest.mark.gpu
def test_gpushare_top_on_real_gpu(source):
    """gpushare-top against the real device source: renders without error,
    shows the MI355X VRAM total."""
    import io

    from gpushare_amd.cli import top

    out = io.StringIO()
    assert top.main([], source=source, out=out) == 0
    text = out.getvalue()
    assert "GiB" in text and "renderD" in text


@pytest.mark.gpu
def test_numa_topology_advertised_from_real_kfd(source, tmp_path):
    """--numa-topology on the real box: every grain must carry its GPU's
    actual NUMA domain (from the DRM device's numa_node) through the
    modern Device.topology field."""
    from gpushare_amd import consts
    from gpushare_amd.allocator import Allocator
    from gpushare_amd.cluster.kubeclient import FakeKubeClient
    from gpushare_amd.cluster.podmanager import PodManager
    from gpushare_amd.deviceplugin import v1beta1 as api
    from gpushare_amd.deviceplugin.server import GPUSharePlugin

    gpus = source.devices()
    kube = FakeKubeClient(node_name="gpu-node")
    pm = PodManager(
        kube, "gpu-node", kubelet_client=kube.as_kubelet(), cache_ttl=0.0
    )
    plugin = GPUSharePlugin(
        gpus,
        Allocator(gpus, pm),
        socket_dir=str(tmp_path),
        numa_topology=True,
    )
    resp = api.ListAndWatchResponse.FromString(plugin.encoded_device_list())
    assert len(resp.devices) == sum(g.mem_units(consts.GIB) for g in gpus)
    for d in resp.devices:
        gpu = gpus[plugin.table.gpu_of[d.ID]]
        if gpu.numa_node >= 0:
            assert [n.ID for n in d.topology.nodes] == [gpu.numa_node]
        else:
            assert not d.HasField("topology")
