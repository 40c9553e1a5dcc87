"""In-process stub kubelet — the test/bench harness the reference never had.

Implements the kubelet side of the device-plugin contract (the
rebuild needs "a stub kubelet (in-process gRPC server implementing
Registration and calling Allocate)"):

- serves ``Registration.Register`` on ``kubelet.sock`` in a chosen dir;
- on registration, dials the plugin's endpoint, calls
  ``GetDevicePluginOptions``, and consumes the ``ListAndWatch`` stream on a
  background thread (device inventory mirrors what a real kubelet would
  track);
- exposes ``allocate(device_ids)`` / grouped container requests the way the
  kubelet issues them at pod admission.

Used by the unit tests (BASELINE config 1: mock GPU + stub kubelet on CPU)
and by bench.py for all five BASELINE configs.
"""

from __future__ import annotations

import logging
import os
import threading
import time
from concurrent import futures
from typing import Optional

import grpc

from .. import consts
from . import v1beta1 as api

log = logging.getLogger(__name__)


class DevicePluginClient:
    """Kubelet's view of one registered plugin."""

    def __init__(
        self,
        socket_path: str,
        resource_name: str,
        max_receive_mb: int = 4,   # the REAL kubelet's gRPC default — keep
                                   # the harness as strict as production
    ):
        self.socket_path = socket_path
        self.resource_name = resource_name
        self._channel = grpc.insecure_channel(
            f"unix://{socket_path}",
            options=[
                ("grpc.max_receive_message_length", max_receive_mb << 20)
            ],
        )
        grpc.channel_ready_future(self._channel).result(timeout=10)

        self._get_options = self._channel.unary_unary(
            api.METHOD_GET_OPTIONS,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=api.DevicePluginOptions.FromString,
        )
        self._allocate = self._channel.unary_unary(
            api.METHOD_ALLOCATE,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=api.AllocateResponse.FromString,
        )
        self._list_and_watch = self._channel.unary_stream(
            api.METHOD_LIST_AND_WATCH,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=api.ListAndWatchResponse.FromString,
        )

        self.options = self._get_options(api.Empty(), timeout=5)

        # device inventory maintained from the ListAndWatch stream
        self._inv_lock = threading.Condition()
        self.devices: dict[str, str] = {}      # id -> health
        self.updates_received = 0
        self._stream_done = threading.Event()
        self._stream = None
        self._thread = threading.Thread(
            target=self._consume_stream, name="kubelet-law", daemon=True
        )
        self._thread.start()

    def _consume_stream(self):
        try:
            self._stream = self._list_and_watch(api.Empty())
            for resp in self._stream:
                with self._inv_lock:
                    self.devices = {d.ID: d.health for d in resp.devices}
                    self.updates_received += 1
                    self._inv_lock.notify_all()
        except grpc.RpcError as e:
            if e.code() not in (
                grpc.StatusCode.CANCELLED,
                grpc.StatusCode.UNAVAILABLE,
            ):
                log.warning("ListAndWatch stream ended: %s", e)
        finally:
            self._stream_done.set()

    # ------------------------------------------------------------------ #
    def wait_for_devices(self, min_count: int = 1, timeout: float = 10.0) -> dict:
        deadline = time.monotonic() + timeout
        with self._inv_lock:
            while len(self.devices) < min_count:
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    raise TimeoutError(
                        f"no device inventory after {timeout}s "
                        f"(have {len(self.devices)})"
                    )
                self._inv_lock.wait(timeout=remaining)
            return dict(self.devices)

    def wait_for_update(self, min_updates: int, timeout: float = 10.0) -> dict:
        deadline = time.monotonic() + timeout
        with self._inv_lock:
            while self.updates_received < min_updates:
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    raise TimeoutError("no ListAndWatch update")
                self._inv_lock.wait(timeout=remaining)
            return dict(self.devices)

    def healthy_devices(self) -> list[str]:
        with self._inv_lock:
            return [i for i, h in self.devices.items() if h == consts.HEALTHY]

    def allocate(
        self, container_device_ids: list[list[str]], timeout: float = 30.0
    ) -> "api.AllocateResponse":
        """One AllocateRequest with one ContainerAllocateRequest per inner
        list — exactly how kubelet groups a pod's containers."""
        req = api.AllocateRequest()
        for ids in container_device_ids:
            cr = req.container_requests.add()
            cr.devicesIDs.extend(ids)
        return self._allocate(req, timeout=timeout)

    def close(self):
        if self._stream is not None:
            self._stream.cancel()
        self._channel.close()
        self._stream_done.wait(timeout=2)


class StubKubelet:
    """Registration server + plugin-client factory."""

    def __init__(self, socket_dir: str):
        self.socket_dir = socket_dir
        self.socket_path = os.path.join(socket_dir, consts.KUBELET_SOCKET_NAME)
        self._server: Optional[grpc.Server] = None
        self._reg_lock = threading.Condition()
        self.plugins: dict[str, DevicePluginClient] = {}   # resource -> client
        self.register_errors: list[str] = []

    # -- Registration service ------------------------------------------- #
    def _register(self, request, context) -> "api.Empty":
        if request.version != consts.API_VERSION:
            context.abort(
                grpc.StatusCode.INVALID_ARGUMENT,
                f"unsupported API version {request.version}",
            )
        endpoint_path = os.path.join(self.socket_dir, request.endpoint)
        log.info(
            "kubelet: plugin registering resource=%s endpoint=%s",
            request.resource_name,
            request.endpoint,
        )
        try:
            client = DevicePluginClient(endpoint_path, request.resource_name)
        except Exception as e:
            self.register_errors.append(str(e))
            context.abort(grpc.StatusCode.INTERNAL, f"cannot dial plugin: {e}")
        with self._reg_lock:
            old = self.plugins.get(request.resource_name)
            if old is not None:
                old.close()
            self.plugins[request.resource_name] = client
            self._reg_lock.notify_all()
        return api.Empty()

    # -- lifecycle ------------------------------------------------------- #
    def start(self) -> None:
        os.makedirs(self.socket_dir, exist_ok=True)
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)
        handlers = {
            "Register": grpc.unary_unary_rpc_method_handler(
                self._register,
                request_deserializer=api.RegisterRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
        }
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))
        self._server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler(api.REGISTRATION_SERVICE, handlers),)
        )
        self._server.add_insecure_port(f"unix://{self.socket_path}")
        self._server.start()
        log.info("stub kubelet serving on %s", self.socket_path)

    def wait_for_plugin(
        self,
        resource_name: str = consts.RESOURCE_NAME,
        timeout: float = 10.0,
        max_receive_mb: int = 0,   # >0: reconnect with a raised gRPC limit
    ) -> DevicePluginClient:
        deadline = time.monotonic() + timeout
        with self._reg_lock:
            while resource_name not in self.plugins:
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    raise TimeoutError(f"plugin {resource_name} never registered")
                self._reg_lock.wait(timeout=remaining)
            client = self.plugins[resource_name]
        if max_receive_mb > 0:
            bigger = DevicePluginClient(
                client.socket_path, resource_name,
                max_receive_mb=max_receive_mb,
            )
            with self._reg_lock:
                self.plugins[resource_name] = bigger
            client.close()
            return bigger
        return client

    def stop(self) -> None:
        for client in self.plugins.values():
            client.close()
        self.plugins.clear()
        if self._server is not None:
            self._server.stop(grace=1).wait(timeout=5)
            self._server = None
        if os.path.exists(self.socket_path):
            try:
                os.unlink(self.socket_path)
            except OSError:
                pass
