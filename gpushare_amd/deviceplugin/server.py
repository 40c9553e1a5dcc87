"""Device-plugin gRPC server (L3) — serves ``amdgpushare.sock``.

Reference counterpart: pkg/gpu/nvidia/server.go.  Differences that matter:

- ListAndWatch streams **pre-encoded** payloads (native codec,
  gpushare_amd/native/devlist_codec.cpp): the steady-state 2,304-device list
  on an 8×MI355X node is one cached buffer, not a per-send re-marshal
  (SURVEY §7 hard part 2).
- A physical GPU health flip marks **all** of that GPU's fake devices, not
  just one (fixes the reference bug noted at SURVEY §3.3 / server.go:175-184),
  and health *recovery* is supported (reference FIXME at server.go:180:
  Unhealthy was terminal).
- Allocation is delegated to :class:`gpushare_amd.allocator.Allocator`.
"""

from __future__ import annotations

import logging
import os
import threading
import time
from concurrent import futures
from typing import Optional

import grpc

from .. import consts, metrics
from ..device import PhysicalGPU
from ..device.fakedev import FakeDeviceTable, make_codec
from . import v1beta1 as api

log = logging.getLogger(__name__)


class GPUSharePlugin:
    def __init__(
        self,
        gpus: list[PhysicalGPU],
        allocator,
        unit: str = consts.GIB,
        resource_name: str = consts.RESOURCE_NAME,
        socket_dir: str = consts.DEVICE_PLUGIN_PATH,
        socket_name: str = consts.SERVER_SOCK_NAME,
        max_workers: int = 64,  # each ListAndWatch stream pins a worker thread
        allow_oversize_inventory: bool = False,
        numa_topology: bool = False,
    ):
        self.gpus = gpus
        self.allocator = allocator
        self.unit = unit
        self.resource_name = resource_name
        self.socket_dir = socket_dir
        self.socket_name = socket_name
        self.socket_path = os.path.join(socket_dir, socket_name)
        self.max_workers = max_workers

        self.table = FakeDeviceTable.build(gpus, unit)
        # --numa-topology: advertise each grain's NUMA domain via the
        # modern Device.topology field so the kubelet TopologyManager can
        # NUMA-align gpu-mem with cpu/memory (matters on a 2-socket 8-OAM
        # MI355X box: 4 GPUs per socket).  Off by default — pre-1.17
        # kubelets skip the unknown field, but byte-parity with the
        # reference's wire traffic is the default stance.
        self._codec = make_codec(
            self.table.ids,
            numa=self.table.numa_of if numa_topology else None,
        )
        metrics.observe_inventory(len(self.table))
        # a payload over the kubelet's 4 MiB gRPC receive default fails
        # RESOURCE_EXHAUSTED on the kubelet side — at MiB grain one 288 GiB
        # MI355X is already ~8.4 MB.  Fail loudly at startup instead.
        self.inventory_bytes = len(self._codec.encode([]))
        if self.inventory_bytes > consts.KUBELET_GRPC_MAX_MSG:
            msg = (
                f"inventory of {len(self.table)} fake devices encodes to "
                f"{self.inventory_bytes / 1e6:.1f} MB — over the kubelet's "
                f"4 MiB gRPC receive limit; use GiB granularity on "
                f"288 GiB-class GPUs (or a kubelet built with a larger "
                f"limit + allow_oversize_inventory)"
            )
            if not allow_oversize_inventory:
                raise ValueError(msg)
            log.warning("%s — continuing as requested", msg)

        # health state: plugin GPU index -> healthy?
        self._health_lock = threading.Condition()
        self._unhealthy_gpus: set[int] = set()
        self._health_version = 0

        self._server: Optional[grpc.Server] = None
        self._stop_event = threading.Event()

    # ------------------------------------------------------------------ #
    # health
    # ------------------------------------------------------------------ #
    def set_gpu_health(self, gpu_index: Optional[int], healthy: bool) -> None:
        """Flip one GPU (or all, if None — mirrors the reference treating a
        UUID-less event as all-devices-unhealthy, nvidia.go:138-144)."""
        with self._health_lock:
            targets = (
                [g.index for g in self.gpus] if gpu_index is None else [gpu_index]
            )
            changed = False
            for idx in targets:
                if healthy and idx in self._unhealthy_gpus:
                    self._unhealthy_gpus.discard(idx)
                    changed = True
                elif not healthy and idx not in self._unhealthy_gpus:
                    self._unhealthy_gpus.add(idx)
                    changed = True
            if changed:
                self._health_version += 1
                self._health_lock.notify_all()
                metrics.observe_health_event(
                    "recovered" if healthy else "unhealthy",
                    len(self._unhealthy_gpus),
                )

    def _unhealthy_fake_indices(self) -> list[int]:
        out: list[int] = []
        for gpu_idx in self._unhealthy_gpus:
            out.extend(self.table.gpu_fake_indices(gpu_idx))
        return out

    def encoded_device_list(self) -> bytes:
        return self._codec.encode(self._unhealthy_fake_indices())

    # ------------------------------------------------------------------ #
    # RPC behaviors
    # ------------------------------------------------------------------ #
    def _get_options(self, request, context) -> "api.DevicePluginOptions":
        return api.DevicePluginOptions(
            pre_start_required=False,
            get_preferred_allocation_available=True,
        )

    def _list_and_watch(self, request, context):
        """Stream: full list once, then re-send on every health change.
        Yields pre-encoded bytes (response_serializer is identity)."""
        with self._health_lock:
            version = self._health_version
            payload = self.encoded_device_list()
        log.info(
            "ListAndWatch: sending %d fake devices (%d bytes)",
            len(self.table),
            len(payload),
        )
        metrics.observe_law_send()
        yield payload
        while not self._stop_event.is_set() and context.is_active():
            with self._health_lock:
                if self._health_version == version:
                    self._health_lock.wait(timeout=1.0)
                if self._health_version == version:
                    continue
                version = self._health_version
                payload = self.encoded_device_list()
                unhealthy = len(self._unhealthy_fake_indices())
            log.info(
                "ListAndWatch: health change -> resend (%d unhealthy grains)",
                unhealthy,
            )
            metrics.observe_law_send()
            yield payload

    def _allocate(self, request, context) -> "api.AllocateResponse":
        return self.allocator.allocate(request)

    def _get_preferred_allocation(
        self, request, context
    ) -> "api.PreferredAllocationResponse":
        """Modern kubelet (>= 1.19) asks which grains to hand Allocate.

        Grains are equivalent WITHIN a GPU but not across GPUs: an
        arbitrary kubelet pick can strand availability across several
        GPUs while the extender packs by whole-GPU memory.  Answer
        binpack-consistently: after honoring must_include, fill from the
        single GPU that can satisfy the remainder with the FEWEST
        available grains (tightest fit), falling back to most-available
        order across GPUs for oversize requests."""
        resp = api.PreferredAllocationResponse()
        for cr in request.container_requests:
            out = resp.container_responses.add()
            chosen = list(cr.must_include_deviceIDs)
            need = cr.allocation_size - len(chosen)
            if need < 0:
                out.deviceIDs.extend(chosen[: cr.allocation_size])
                continue
            chosen_set = set(chosen)
            by_gpu: dict[int, list[str]] = {}
            for dev_id in cr.available_deviceIDs:
                if dev_id in chosen_set:
                    continue
                gpu_idx = self.table.gpu_of.get(dev_id)
                if gpu_idx is not None:
                    by_gpu.setdefault(gpu_idx, []).append(dev_id)
            # tightest single GPU first (binpack), most-available fallback
            fitting = sorted(
                (ids for ids in by_gpu.values() if len(ids) >= need),
                key=len,
            )
            if need and fitting:
                chosen.extend(fitting[0][:need])
            else:
                for ids in sorted(by_gpu.values(), key=len, reverse=True):
                    if need <= 0:
                        break
                    take = ids[:need]
                    chosen.extend(take)
                    need -= len(take)
            out.deviceIDs.extend(chosen)
        return resp

    def _pre_start(self, request, context) -> "api.PreStartContainerResponse":
        return api.PreStartContainerResponse()

    # ------------------------------------------------------------------ #
    # lifecycle
    # ------------------------------------------------------------------ #
    def start(self) -> None:
        """Listen on our unix socket and verify it accepts connections
        (reference: server.go:106-134 self-dial check)."""
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)
        os.makedirs(self.socket_dir, exist_ok=True)

        handlers = {
            "GetDevicePluginOptions": grpc.unary_unary_rpc_method_handler(
                self._get_options,
                request_deserializer=api.Empty.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
            "ListAndWatch": grpc.unary_stream_rpc_method_handler(
                self._list_and_watch,
                request_deserializer=api.Empty.FromString,
                response_serializer=None,  # pre-encoded bytes pass through
            ),
            "Allocate": grpc.unary_unary_rpc_method_handler(
                self._allocate,
                request_deserializer=api.AllocateRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
            "PreStartContainer": grpc.unary_unary_rpc_method_handler(
                self._pre_start,
                request_deserializer=api.PreStartContainerRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
            "GetPreferredAllocation": grpc.unary_unary_rpc_method_handler(
                self._get_preferred_allocation,
                request_deserializer=(
                    api.PreferredAllocationRequest.FromString
                ),
                response_serializer=lambda m: m.SerializeToString(),
            ),
        }
        self._server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=self.max_workers)
        )
        self._server.add_generic_rpc_handlers(
            (
                grpc.method_handlers_generic_handler(
                    api.DEVICEPLUGIN_SERVICE, handlers
                ),
            )
        )
        self._server.add_insecure_port(f"unix://{self.socket_path}")
        self._stop_event.clear()
        self._server.start()

        # self-dial liveness check
        with grpc.insecure_channel(f"unix://{self.socket_path}") as ch:
            grpc.channel_ready_future(ch).result(timeout=5)
        log.info("device plugin serving on %s", self.socket_path)

    def register(self, kubelet_socket: Optional[str] = None) -> None:
        """Register with the kubelet (reference: server.go:150-169)."""
        if kubelet_socket is None:
            kubelet_socket = os.path.join(
                self.socket_dir, consts.KUBELET_SOCKET_NAME
            )
        req = api.RegisterRequest(
            version=consts.API_VERSION,
            endpoint=self.socket_name,
            resource_name=self.resource_name,
        )
        with grpc.insecure_channel(f"unix://{kubelet_socket}") as ch:
            grpc.channel_ready_future(ch).result(timeout=10)
            register = ch.unary_unary(
                api.METHOD_REGISTER,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=api.Empty.FromString,
            )
            register(req, timeout=10)
        log.info(
            "registered resource %s (%d fake devices) with kubelet at %s",
            self.resource_name,
            len(self.table),
            kubelet_socket,
        )

    def serve(self, kubelet_socket: Optional[str] = None) -> None:
        self.start()
        self.register(kubelet_socket)

    def stop(self) -> None:
        """Graceful stop (reference: server.go:137-147)."""
        self._stop_event.set()
        with self._health_lock:
            self._health_lock.notify_all()
        if self._server is not None:
            self._server.stop(grace=1).wait(timeout=5)
            self._server = None
        if os.path.exists(self.socket_path):
            try:
                os.unlink(self.socket_path)
            except OSError:
                pass
        if hasattr(self.allocator, "close"):
            self.allocator.close()

    # convenience for tests
    def wait_stopped(self, timeout: float = 5.0) -> None:
        deadline = time.monotonic() + timeout
        while self._server is not None and time.monotonic() < deadline:
            time.sleep(0.01)
