"""Device layer: mock source, fake-device expansion, KFD topology walker."""

import pytest

from gpushare_amd import consts
from gpushare_amd.device import PhysicalGPU, create_source
from gpushare_amd.device import fakedev, kfd_topology
from gpushare_amd.device.mock_source import MockSource


def test_mock_spec_parsing():
    src = MockSource.from_spec("8x288GiB")
    gpus = src.devices()
    assert len(gpus) == 8
    assert gpus[0].memory_bytes == 288 << 30
    assert gpus[0].mem_units(consts.GIB) == 288
    assert gpus[0].mem_units(consts.MIB) == 288 * 1024


def test_mock_spec_invalid():
    with pytest.raises(ValueError):
        MockSource.from_spec("eight-gpus")


def test_create_source_env(monkeypatch):
    monkeypatch.setenv("GPUSHARE_MOCK_SPEC", "1x8GiB")
    src = create_source()
    assert len(src.devices()) == 1
    assert src.devices()[0].mem_units(consts.GIB) == 8


def test_fake_id_roundtrip():
    fid = fakedev.fake_id("amd-00ff", 287)
    assert fid == "amd-00ff-_-287"
    assert fakedev.real_id(fid) == "amd-00ff"


def test_fake_id_length_cap():
    with pytest.raises(ValueError):
        fakedev.fake_id("u" * 62, 1000)


def test_expansion_mi355x_node():
    gpus = MockSource.from_spec("8x288GiB").devices()
    table = fakedev.FakeDeviceTable.build(gpus, consts.GIB)
    assert len(table) == 8 * 288
    assert all(len(i) <= consts.MAX_DEVICE_ID_LEN for i in table.ids)
    # GPU-major layout and correct back-mapping
    assert table.gpu_of[table.ids[0]] == 0
    assert table.gpu_of[table.ids[288]] == 1
    assert list(table.gpu_fake_indices(2)) == list(range(2 * 288, 3 * 288))
    assert table.uuid_of[3] == gpus[3].uuid
    assert table.index_of[gpus[3].uuid] == 3


def test_expansion_rejects_zero_memory():
    gpu = PhysicalGPU(index=0, uuid="z", memory_bytes=0)
    with pytest.raises(ValueError):
        fakedev.FakeDeviceTable.build([gpu], consts.GIB)


# --------------------------------------------------------------------------- #
# KFD topology from a fake sysfs tree
# --------------------------------------------------------------------------- #

def _write(p, text):
    p.parent.mkdir(parents=True, exist_ok=True)
    p.write_text(text)


@pytest.fixture
def fake_kfd(tmp_path):
    """2 GPUs + 1 CPU node; xGMI between the GPUs; renderD129/130."""
    root = tmp_path / "nodes"
    # node 0: CPU
    _write(root / "0" / "gpu_id", "0\n")
    _write(root / "0" / "properties", "cpu_cores_count 32\nsimd_count 0\n")
    for n, (gpuid, minor, uid) in enumerate(
        [(1234, 129, 0xAB01), (5678, 130, 0xAB02)], start=1
    ):
        _write(root / str(n) / "gpu_id", f"{gpuid}\n")
        _write(
            root / str(n) / "properties",
            f"simd_count 1024\ngfx_target_version 90500\n"
            f"drm_render_minor {minor}\nunique_id {uid}\n",
        )
        _write(
            root / str(n) / "mem_banks" / "0" / "properties",
            f"heap_type 1\nsize_in_bytes {8 << 30}\n",
        )
        # xGMI link to the other GPU node, PCIe link to CPU node 0
        other = 2 if n == 1 else 1
        _write(
            root / str(n) / "io_links" / "0" / "properties",
            f"type 11\nnode_from {n}\nnode_to {other}\nweight 15\n",
        )
        _write(
            root / str(n) / "io_links" / "1" / "properties",
            f"type 2\nnode_from {n}\nnode_to 0\nweight 20\n",
        )
    return str(root)


def test_kfd_topology_parse(fake_kfd):
    nodes = kfd_topology.read_topology(fake_kfd)
    assert len(nodes) == 3
    gpus = [n for n in nodes if n.is_gpu]
    assert len(gpus) == 2
    assert gpus[0].render_minor == 129
    assert gpus[0].vram_bytes == 8 << 30
    assert gpus[0].xgmi_peer_nodes == [2]
    assert gpus[0].pcie_peer_nodes == [0]


def test_kfd_resolve(fake_kfd, tmp_path):
    topo = kfd_topology.resolve(fake_kfd, drm_root=str(tmp_path / "no-drm"))
    assert set(topo.keys()) == {1234, 5678}
    t = topo[1234]
    assert t.render_path == "/dev/dri/renderD129"
    assert t.unique_id == 0xAB01
    assert t.xgmi_peer_gpu_ids == [5678]
    assert t.vram_bytes == 8 << 30
    assert t.gfx_target_version == 90500


def test_mock_heterogeneous_spec():
    """'288+288+96GiB' — per-GPU capacities differ (the reference assumes a
    homogeneous node, nvidia.go:70-72; we advertise real per-GPU sizes)."""
    from gpushare_amd.device.mock_source import MockSource

    gpus = MockSource.from_spec("288+288+96GiB").devices()
    assert [g.mem_units("GiB") for g in gpus] == [288, 288, 96]
    assert gpus[2].xgmi_peers == (0, 1)


def test_memguard_lib_loads_and_reads_env(tmp_path):
    """CPU-side smoke of libgpushare_memguard.so: the lib must dlopen
    anywhere (links only libdl/libc), parse the env budget, and map its
    container-scoped slot table (no HIP runtime involved)."""
    import os
    import subprocess
    import sys

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    if not os.path.exists(lib):
        import pytest

        pytest.skip("memguard not built")
    code = (
        "import ctypes, os\n"
        f"l = ctypes.CDLL({lib!r})\n"
        "l.gpushare_memguard_limit.restype = ctypes.c_int64\n"
        "l.gpushare_memguard_used.restype = ctypes.c_int64\n"
        "assert l.gpushare_memguard_limit() == 1 << 30\n"
        "assert l.gpushare_memguard_used() == 0\n"
        "print('CPU_MEMGUARD_OK')\n"
    )
    env = dict(os.environ)
    env["GPUSHARE_MEM_LIMIT_BYTES"] = str(1 << 30)
    env["GPUSHARE_POD_UID"] = f"cputest-{os.getpid()}"
    out = subprocess.run(
        [sys.executable, "-c", code], env=env, capture_output=True, text=True
    )
    shm = f"/dev/shm/gpushare.memguard.cputest-{os.getpid()}.c"
    if os.path.exists(shm):
        os.unlink(shm)
    assert "CPU_MEMGUARD_OK" in out.stdout, out.stderr


def test_memguard_container_token_scopes_table():
    """Containers of one pod share /dev/shm but not a PID namespace, so
    each container must get its OWN accounting table: the
    GPUSHARE_CONTAINER_TOKEN env is part of the table path."""
    import os
    import subprocess
    import sys

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    if not os.path.exists(lib):
        import pytest

        pytest.skip("memguard not built")
    uid = f"toktest-{os.getpid()}"
    code = (
        "import ctypes\n"
        f"l = ctypes.CDLL({lib!r})\n"
        "l.gpushare_memguard_used.restype = ctypes.c_int64\n"
        "assert l.gpushare_memguard_used() == 0\n"
    )
    paths = []
    try:
        for token in ("c0", "c1"):
            env = dict(os.environ)
            env["GPUSHARE_MEM_LIMIT_BYTES"] = str(1 << 30)
            env["GPUSHARE_POD_UID"] = uid
            env["GPUSHARE_CONTAINER_TOKEN"] = token
            out = subprocess.run(
                [sys.executable, "-c", code],
                env=env,
                capture_output=True,
                text=True,
            )
            assert out.returncode == 0, out.stderr
            paths.append(f"/dev/shm/gpushare.memguard.{uid}.{token}")
        # both containers created distinct tables
        for p in paths:
            assert os.path.exists(p), p
    finally:
        for p in paths:
            if os.path.exists(p):
                os.unlink(p)


def test_memguard_per_device_limits_parsed():
    """GPUSHARE_MEM_LIMIT_BYTES_PER_DEVICE drives per-ordinal caps; absent
    entries stay uncapped (-1)."""
    import os
    import subprocess
    import sys

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    if not os.path.exists(lib):
        import pytest

        pytest.skip("memguard not built")
    uid = f"devtest-{os.getpid()}"
    code = (
        "import ctypes\n"
        f"l = ctypes.CDLL({lib!r})\n"
        "l.gpushare_memguard_dev_limit.restype = ctypes.c_int64\n"
        "l.gpushare_memguard_dev_limit.argtypes = [ctypes.c_int]\n"
        "assert l.gpushare_memguard_dev_limit(0) == 1 << 30, l.gpushare_memguard_dev_limit(0)\n"
        "assert l.gpushare_memguard_dev_limit(1) == 2 << 30\n"
        "assert l.gpushare_memguard_dev_limit(2) == -1\n"
        "print('PER_DEVICE_OK')\n"
    )
    env = dict(os.environ)
    env["GPUSHARE_MEM_LIMIT_BYTES"] = str(3 << 30)
    env["GPUSHARE_MEM_LIMIT_BYTES_PER_DEVICE"] = f"{1 << 30},{2 << 30}"
    env["GPUSHARE_POD_UID"] = uid
    out = subprocess.run(
        [sys.executable, "-c", code], env=env, capture_output=True, text=True
    )
    shm = f"/dev/shm/gpushare.memguard.{uid}.c"
    if os.path.exists(shm):
        os.unlink(shm)
    assert "PER_DEVICE_OK" in out.stdout, out.stderr


def test_memguard_fork_child_gets_own_slot():
    """After fork() each process must own its own slot (a child never
    charges or repays its parent's reservation): two live pids appear in
    the container's table.  Slot layout (memguard.cpp struct Slot):
    pid(i32) pad(i32) used(i64) used_dev[8](i64) = 80 bytes."""
    import os
    import struct
    import subprocess
    import sys

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    if not os.path.exists(lib):
        import pytest

        pytest.skip("memguard not built")
    uid = f"forktest-{os.getpid()}"
    code = f"""
import ctypes, os, sys, time
l = ctypes.CDLL({lib!r})
l.gpushare_memguard_used.restype = ctypes.c_int64
assert l.gpushare_memguard_used() == 0      # parent binds its slot
pid = os.fork()
if pid == 0:
    assert l.gpushare_memguard_used() == 0  # child binds a NEW slot
    time.sleep(30)                          # hold the slot until killed
    os._exit(0)
time.sleep(0.5)                             # let the child bind
print("PARENT", os.getpid(), "CHILD", pid, flush=True)
sys.stdout.flush()
time.sleep(5)
os.kill(pid, 9)
"""
    env = dict(os.environ)
    env["GPUSHARE_MEM_LIMIT_BYTES"] = str(1 << 30)
    env["GPUSHARE_POD_UID"] = uid
    shm = f"/dev/shm/gpushare.memguard.{uid}.c"
    proc = subprocess.Popen(
        [sys.executable, "-c", code], env=env,
        stdout=subprocess.PIPE, text=True,
    )
    try:
        line = proc.stdout.readline()
        assert line.startswith("PARENT"), line
        _, ppid, _, cpid = line.split()
        with open(shm, "rb") as fh:
            table = fh.read()
        slot_size = 80
        pids = set()
        for off in range(0, len(table), slot_size):
            (pid,) = struct.unpack_from("<i", table, off)
            if pid:
                pids.add(pid)
        assert pids == {int(ppid), int(cpid)}, (
            f"expected parent+child slots, got {pids}"
        )
    finally:
        proc.kill()
        proc.wait(timeout=10)
        if os.path.exists(shm):
            os.unlink(shm)


def test_memguard_fork_clears_inherited_tracking():
    """A forked child inherits the parent's address space but NOT its GPU
    allocations: freeing an inherited pointer in the child must be a
    no-op for the child's budget (tracking map cleared at fork), so the
    child's slot can never go negative."""
    import os
    import struct
    import subprocess
    import sys

    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    if not os.path.exists(lib):
        import pytest

        pytest.skip("memguard not built")
    uid = f"atfork-{os.getpid()}"
    # exercise via the introspection API: there is no GPU here, so drive
    # reserve/track indirectly by checking used() stays 0 in the child
    # and the child owns its own zeroed slot
    code = f"""
import ctypes, os, sys, time
l = ctypes.CDLL({lib!r})
l.gpushare_memguard_used.restype = ctypes.c_int64
assert l.gpushare_memguard_used() == 0
pid = os.fork()
if pid == 0:
    # child: fresh slot, zero usage, and allocations tracked by the
    # parent are invisible (map cleared by the atfork handler)
    assert l.gpushare_memguard_used() == 0
    os._exit(0)
_, status = os.waitpid(pid, 0)
sys.exit(os.waitstatus_to_exitcode(status))
"""
    env = dict(os.environ)
    env["GPUSHARE_MEM_LIMIT_BYTES"] = str(1 << 30)
    env["GPUSHARE_POD_UID"] = uid
    out = subprocess.run(
        [sys.executable, "-c", code], env=env, capture_output=True, text=True
    )
    shm = f"/dev/shm/gpushare.memguard.{uid}.c"
    if os.path.exists(shm):
        os.unlink(shm)
    assert out.returncode == 0, out.stderr[-800:]
