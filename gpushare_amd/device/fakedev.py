"""Fake-device expansion: 1 memory grain = 1 schedulable device.

Reference semantics (pkg/gpu/nvidia/nvidia.go:26-32,53-89): a GPU with M
memory units is advertised as M devices with IDs ``<uuid>-_-<j>``; the real
device is recovered by splitting on ``-_-``.  On MI355X at GiB granularity
that is 288 fake devices per GPU / 2,304 per 8-GPU node, so the expanded
table and its encodings are built once and reused (native codec:
gpushare_amd/native/devlist_codec.cpp).
"""

from __future__ import annotations

from dataclasses import dataclass

from .. import consts
from . import PhysicalGPU

SEPARATOR = "-_-"


def fake_id(uuid: str, j: int) -> str:
    fid = f"{uuid}{SEPARATOR}{j}"
    if len(fid) > consts.MAX_DEVICE_ID_LEN:
        raise ValueError(
            f"fake device ID {fid!r} exceeds {consts.MAX_DEVICE_ID_LEN} chars "
            "(api.proto:84); shorten the GPU uuid"
        )
    return fid


def real_id(fake: str) -> str:
    """Reference: extractRealDeviceID (nvidia.go:30-32)."""
    return fake.split(SEPARATOR)[0]


@dataclass
class FakeDeviceTable:
    """Immutable expansion of physical GPUs into fake devices."""

    ids: list[str]                 # all fake device IDs, GPU-major order
    gpu_of: dict[str, int]         # fake id -> plugin GPU index
    uuid_of: dict[int, str]        # GPU index -> uuid
    index_of: dict[str, int]       # uuid -> GPU index
    ranges: dict[int, tuple[int, int]]  # GPU index -> [start, end) into ids
    unit: str
    units_per_gpu: list[int]
    numa_of: list[int]             # fake idx -> NUMA node (-1 unknown)

    @classmethod
    def build(cls, gpus: list[PhysicalGPU], unit: str) -> "FakeDeviceTable":
        if unit not in consts.VALID_MEMORY_UNITS:
            raise ValueError(f"invalid memory unit {unit!r}")
        ids: list[str] = []
        gpu_of: dict[str, int] = {}
        ranges: dict[int, tuple[int, int]] = {}
        units_per_gpu: list[int] = []
        numa_of: list[int] = []
        for g in gpus:
            n = g.mem_units(unit)
            if n <= 0:
                raise ValueError(f"GPU {g.index} reports no memory")
            start = len(ids)
            numa = getattr(g, "numa_node", -1)
            for j in range(n):
                fid = fake_id(g.uuid, j)
                ids.append(fid)
                gpu_of[fid] = g.index
                numa_of.append(numa if numa is not None else -1)
            # keyed by the PLUGIN index, not list position — robust to
            # non-contiguous index sets (e.g. partitioned/filtered nodes)
            ranges[g.index] = (start, len(ids))
            units_per_gpu.append(n)
        return cls(
            ids=ids,
            gpu_of=gpu_of,
            uuid_of={g.index: g.uuid for g in gpus},
            index_of={g.uuid: g.index for g in gpus},
            ranges=ranges,
            unit=unit,
            units_per_gpu=units_per_gpu,
            numa_of=numa_of,
        )

    def __len__(self) -> int:
        return len(self.ids)

    def gpu_fake_indices(self, gpu_index: int) -> range:
        """All positions in ``ids`` belonging to one physical GPU — used to
        flip *every* grain of a failed GPU (fixes the reference bug where a
        health event flips a single fake device, server.go:175-184)."""
        start, end = self.ranges[gpu_index]
        return range(start, end)


def encode_list_python(
    ids: list[str], unhealthy: set[int], numa: list[int] | None = None
) -> bytes:
    """Pure-python ListAndWatchResponse encoder — fallback and test oracle
    for the native codec; byte-identical output.  ``numa`` (optional,
    per fake device, -1 = omit) adds Device.topology NUMA hints — the
    modern kubelet TopologyManager field (upstream api.proto field 3)."""

    def varint(v: int) -> bytes:
        out = bytearray()
        while v >= 0x80:
            out.append((v & 0x7F) | 0x80)
            v >>= 7
        out.append(v)
        return bytes(out)

    out = bytearray()
    for k, fid in enumerate(ids):
        health = b"Unhealthy" if k in unhealthy else b"Healthy"
        idb = fid.encode()
        dev = b"\x0a" + varint(len(idb)) + idb + b"\x12" + varint(len(health)) + health
        if numa is not None and numa[k] >= 0:
            # NUMANode{ID=<n>} wrapped in TopologyInfo.nodes wrapped in
            # Device.topology (field 3, wire type 2)
            numanode = b"\x08" + varint(numa[k])
            topo = b"\x0a" + varint(len(numanode)) + numanode
            dev += b"\x1a" + varint(len(topo)) + topo
        out += b"\x0a" + varint(len(dev)) + dev
    return bytes(out)


def make_codec(ids: list[str], numa: list[int] | None = None):
    """Native codec if built, else a python shim with the same interface.
    ``numa``: optional per-device NUMA node (-1 = omit topology)."""
    try:
        from .. import _devlist

        if numa is not None:
            return _devlist.DeviceListCodec(ids, numa)
        return _devlist.DeviceListCodec(ids)
    except ImportError:

        class _PyCodec:
            def __init__(self, ids_, numa_):
                self._ids = list(ids_)
                self._numa = list(numa_) if numa_ is not None else None

            def __len__(self):
                return len(self._ids)

            def encode(self, unhealthy=()):
                return encode_list_python(
                    self._ids, set(unhealthy), self._numa
                )

        return _PyCodec(ids, numa)
