"""Kubelet device-plugin API v1beta1 message classes, built at import time.

The kubelet device-plugin gRPC contract (two services: ``Registration`` on the
kubelet's socket and ``DevicePlugin`` on ours) is an external wire protocol we
must speak exactly (reference contract:
vendor/k8s.io/kubernetes/pkg/kubelet/apis/deviceplugin/v1beta1/api.proto:23-161).

There is no protoc/grpcio-tools in the target environment, so instead of
checked-in generated code we construct the ``FileDescriptorProto``
programmatically and get real protobuf message classes from the runtime.
Field numbers/types below ARE the contract — do not renumber.
"""

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_PKG = "v1beta1"

_LABEL_OPTIONAL = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
_LABEL_REPEATED = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
_TYPE_STRING = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
_TYPE_BOOL = descriptor_pb2.FieldDescriptorProto.TYPE_BOOL
_TYPE_INT64 = descriptor_pb2.FieldDescriptorProto.TYPE_INT64
_TYPE_MESSAGE = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE


def _field(name, number, ftype, label=_LABEL_OPTIONAL, type_name=None):
    f = descriptor_pb2.FieldDescriptorProto(
        name=name, number=number, type=ftype, label=label
    )
    if type_name:
        f.type_name = type_name
    return f


def _msg(name, *fields, nested=()):
    m = descriptor_pb2.DescriptorProto(name=name)
    m.field.extend(fields)
    m.nested_type.extend(nested)
    return m


def _map_entry(name):
    """A map<string,string> entry message (proto3 map field encoding)."""
    e = _msg(
        name,
        _field("key", 1, _TYPE_STRING),
        _field("value", 2, _TYPE_STRING),
    )
    e.options.map_entry = True
    return e


def _build_file() -> descriptor_pb2.FileDescriptorProto:
    f = descriptor_pb2.FileDescriptorProto(
        name="gpushare_amd/deviceplugin/api.proto",
        package=_PKG,
        syntax="proto3",
    )
    t = lambda n: f".{_PKG}.{n}"  # noqa: E731

    f.message_type.extend(
        [
            # pre_start_required is the vendored-era field; field 2 and the
            # GetPreferredAllocation messages below are the MODERN upstream
            # additions (k8s >= 1.19 api.proto) — a pre-1.19 kubelet skips
            # the unknown bool and never calls the RPC.
            _msg(
                "DevicePluginOptions",
                _field("pre_start_required", 1, _TYPE_BOOL),
                _field("get_preferred_allocation_available", 2, _TYPE_BOOL),
            ),
            _msg(
                "RegisterRequest",
                _field("version", 1, _TYPE_STRING),
                _field("endpoint", 2, _TYPE_STRING),
                _field("resource_name", 3, _TYPE_STRING),
                _field(
                    "options", 4, _TYPE_MESSAGE, type_name=t("DevicePluginOptions")
                ),
            ),
            _msg("Empty"),
            _msg(
                "ListAndWatchResponse",
                _field(
                    "devices", 1, _TYPE_MESSAGE, _LABEL_REPEATED, t("Device")
                ),
            ),
            # Device.topology (field 3) + TopologyInfo/NUMANode are the
            # MODERN upstream additions (k8s ≥1.17 api.proto; the vendored
            # snapshot predates them): NUMA hints for the kubelet
            # TopologyManager.  Field numbers match upstream exactly; a
            # pre-1.17 kubelet skips field 3 as an unknown (proto3), so
            # sending it is always safe — the plugin still gates it
            # behind --numa-topology for byte-parity-by-default.
            _msg(
                "Device",
                _field("ID", 1, _TYPE_STRING),
                _field("health", 2, _TYPE_STRING),
                _field("topology", 3, _TYPE_MESSAGE, type_name=t("TopologyInfo")),
            ),
            _msg(
                "TopologyInfo",
                _field("nodes", 1, _TYPE_MESSAGE, _LABEL_REPEATED, t("NUMANode")),
            ),
            _msg(
                "NUMANode",
                _field("ID", 1, _TYPE_INT64),
            ),
            _msg(
                "PreStartContainerRequest",
                _field("devicesIDs", 1, _TYPE_STRING, _LABEL_REPEATED),
            ),
            _msg("PreStartContainerResponse"),
            _msg(
                "AllocateRequest",
                _field(
                    "container_requests",
                    1,
                    _TYPE_MESSAGE,
                    _LABEL_REPEATED,
                    t("ContainerAllocateRequest"),
                ),
            ),
            _msg(
                "ContainerAllocateRequest",
                _field("devicesIDs", 1, _TYPE_STRING, _LABEL_REPEATED),
            ),
            _msg(
                "AllocateResponse",
                _field(
                    "container_responses",
                    1,
                    _TYPE_MESSAGE,
                    _LABEL_REPEATED,
                    t("ContainerAllocateResponse"),
                ),
            ),
            _msg(
                "ContainerAllocateResponse",
                _field(
                    "envs",
                    1,
                    _TYPE_MESSAGE,
                    _LABEL_REPEATED,
                    t("ContainerAllocateResponse.EnvsEntry"),
                ),
                _field("mounts", 2, _TYPE_MESSAGE, _LABEL_REPEATED, t("Mount")),
                _field(
                    "devices", 3, _TYPE_MESSAGE, _LABEL_REPEATED, t("DeviceSpec")
                ),
                _field(
                    "annotations",
                    4,
                    _TYPE_MESSAGE,
                    _LABEL_REPEATED,
                    t("ContainerAllocateResponse.AnnotationsEntry"),
                ),
                nested=(_map_entry("EnvsEntry"), _map_entry("AnnotationsEntry")),
            ),
            _msg(
                "PreferredAllocationRequest",
                _field(
                    "container_requests",
                    1,
                    _TYPE_MESSAGE,
                    _LABEL_REPEATED,
                    t("ContainerPreferredAllocationRequest"),
                ),
            ),
            _msg(
                "ContainerPreferredAllocationRequest",
                _field("available_deviceIDs", 1, _TYPE_STRING, _LABEL_REPEATED),
                _field(
                    "must_include_deviceIDs", 2, _TYPE_STRING, _LABEL_REPEATED
                ),
                _field(
                    "allocation_size",
                    3,
                    descriptor_pb2.FieldDescriptorProto.TYPE_INT32,
                ),
            ),
            _msg(
                "PreferredAllocationResponse",
                _field(
                    "container_responses",
                    1,
                    _TYPE_MESSAGE,
                    _LABEL_REPEATED,
                    t("ContainerPreferredAllocationResponse"),
                ),
            ),
            _msg(
                "ContainerPreferredAllocationResponse",
                _field("deviceIDs", 1, _TYPE_STRING, _LABEL_REPEATED),
            ),
            _msg(
                "Mount",
                _field("container_path", 1, _TYPE_STRING),
                _field("host_path", 2, _TYPE_STRING),
                _field("read_only", 3, _TYPE_BOOL),
            ),
            _msg(
                "DeviceSpec",
                _field("container_path", 1, _TYPE_STRING),
                _field("host_path", 2, _TYPE_STRING),
                _field("permissions", 3, _TYPE_STRING),
            ),
        ]
    )
    return f


_pool = descriptor_pool.DescriptorPool()
_pool.Add(_build_file())


def _cls(name):
    return message_factory.GetMessageClass(
        _pool.FindMessageTypeByName(f"{_PKG}.{name}")
    )


DevicePluginOptions = _cls("DevicePluginOptions")
RegisterRequest = _cls("RegisterRequest")
Empty = _cls("Empty")
ListAndWatchResponse = _cls("ListAndWatchResponse")
Device = _cls("Device")
TopologyInfo = _cls("TopologyInfo")
NUMANode = _cls("NUMANode")
PreStartContainerRequest = _cls("PreStartContainerRequest")
PreStartContainerResponse = _cls("PreStartContainerResponse")
AllocateRequest = _cls("AllocateRequest")
ContainerAllocateRequest = _cls("ContainerAllocateRequest")
PreferredAllocationRequest = _cls("PreferredAllocationRequest")
ContainerPreferredAllocationRequest = _cls(
    "ContainerPreferredAllocationRequest"
)
PreferredAllocationResponse = _cls("PreferredAllocationResponse")
ContainerPreferredAllocationResponse = _cls(
    "ContainerPreferredAllocationResponse"
)
AllocateResponse = _cls("AllocateResponse")
ContainerAllocateResponse = _cls("ContainerAllocateResponse")
Mount = _cls("Mount")
DeviceSpec = _cls("DeviceSpec")

# gRPC method full names (package + service are part of the wire contract).
REGISTRATION_SERVICE = f"{_PKG}.Registration"
DEVICEPLUGIN_SERVICE = f"{_PKG}.DevicePlugin"

METHOD_REGISTER = f"/{REGISTRATION_SERVICE}/Register"
METHOD_GET_OPTIONS = f"/{DEVICEPLUGIN_SERVICE}/GetDevicePluginOptions"
METHOD_LIST_AND_WATCH = f"/{DEVICEPLUGIN_SERVICE}/ListAndWatch"
METHOD_ALLOCATE = f"/{DEVICEPLUGIN_SERVICE}/Allocate"
METHOD_PRE_START = f"/{DEVICEPLUGIN_SERVICE}/PreStartContainer"
METHOD_GET_PREFERRED = f"/{DEVICEPLUGIN_SERVICE}/GetPreferredAllocation"
