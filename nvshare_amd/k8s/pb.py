"""Kubelet device-plugin API v1beta1 protobuf messages, built at
runtime with descriptor_pb2 (no protoc / grpc_tools in this image).

Field numbers and wire types mirror the upstream
k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto, which is what
the reference's Go plugin compiled against (reference
kubernetes/device-plugin/{server,devices}.go).
"""

from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto

_pool = descriptor_pool.DescriptorPool()


def _msg(fdp, name):
    m = fdp.message_type.add()
    m.name = name
    return m


def _field(m, name, number, ftype, label=_F.LABEL_OPTIONAL,
           type_name=None):
    f = m.field.add()
    f.name = name
    f.number = number
    f.type = ftype
    f.label = label
    if type_name:
        f.type_name = type_name
    return f


def _map_field(fdp, m, name, number):
    """Add a map<string,string> field (nested MapEntry message)."""
    entry = m.nested_type.add()
    entry.name = name.capitalize() + "Entry"
    entry.options.map_entry = True
    _field(entry, "key", 1, _F.TYPE_STRING)
    _field(entry, "value", 2, _F.TYPE_STRING)
    _field(m, name, number, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           f".v1beta1.{m.name}.{entry.name}")


def _build() -> dict:
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "deviceplugin_v1beta1.proto"
    fdp.package = "v1beta1"
    fdp.syntax = "proto3"

    _msg(fdp, "Empty")

    m = _msg(fdp, "DevicePluginOptions")
    _field(m, "pre_start_required", 1, _F.TYPE_BOOL)
    _field(m, "get_preferred_allocation_available", 2, _F.TYPE_BOOL)

    m = _msg(fdp, "RegisterRequest")
    _field(m, "version", 1, _F.TYPE_STRING)
    _field(m, "endpoint", 2, _F.TYPE_STRING)
    _field(m, "resource_name", 3, _F.TYPE_STRING)
    _field(m, "options", 4, _F.TYPE_MESSAGE,
           type_name=".v1beta1.DevicePluginOptions")

    m = _msg(fdp, "NUMANode")
    _field(m, "ID", 1, _F.TYPE_INT64)

    m = _msg(fdp, "TopologyInfo")
    _field(m, "nodes", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           ".v1beta1.NUMANode")

    m = _msg(fdp, "Device")
    _field(m, "ID", 1, _F.TYPE_STRING)
    _field(m, "health", 2, _F.TYPE_STRING)
    _field(m, "topology", 3, _F.TYPE_MESSAGE,
           type_name=".v1beta1.TopologyInfo")

    m = _msg(fdp, "ListAndWatchResponse")
    _field(m, "devices", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           ".v1beta1.Device")

    m = _msg(fdp, "ContainerAllocateRequest")
    _field(m, "devicesIDs", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)

    m = _msg(fdp, "AllocateRequest")
    _field(m, "container_requests", 1, _F.TYPE_MESSAGE,
           _F.LABEL_REPEATED, ".v1beta1.ContainerAllocateRequest")

    m = _msg(fdp, "Mount")
    _field(m, "container_path", 1, _F.TYPE_STRING)
    _field(m, "host_path", 2, _F.TYPE_STRING)
    _field(m, "read_only", 3, _F.TYPE_BOOL)

    m = _msg(fdp, "DeviceSpec")
    _field(m, "container_path", 1, _F.TYPE_STRING)
    _field(m, "host_path", 2, _F.TYPE_STRING)
    _field(m, "permissions", 3, _F.TYPE_STRING)

    m = _msg(fdp, "ContainerAllocateResponse")
    _map_field(fdp, m, "envs", 1)
    _field(m, "mounts", 2, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           ".v1beta1.Mount")
    _field(m, "devices", 3, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           ".v1beta1.DeviceSpec")
    _map_field(fdp, m, "annotations", 4)

    m = _msg(fdp, "AllocateResponse")
    _field(m, "container_responses", 1, _F.TYPE_MESSAGE,
           _F.LABEL_REPEATED, ".v1beta1.ContainerAllocateResponse")

    m = _msg(fdp, "PreStartContainerRequest")
    _field(m, "devicesIDs", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)

    _msg(fdp, "PreStartContainerResponse")

    m = _msg(fdp, "ContainerPreferredAllocationRequest")
    _field(m, "available_deviceIDs", 1, _F.TYPE_STRING,
           _F.LABEL_REPEATED)
    _field(m, "must_include_deviceIDs", 2, _F.TYPE_STRING,
           _F.LABEL_REPEATED)
    _field(m, "allocation_size", 3, _F.TYPE_INT32)

    m = _msg(fdp, "PreferredAllocationRequest")
    _field(m, "container_requests", 1, _F.TYPE_MESSAGE,
           _F.LABEL_REPEATED,
           ".v1beta1.ContainerPreferredAllocationRequest")

    m = _msg(fdp, "ContainerPreferredAllocationResponse")
    _field(m, "deviceIDs", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)

    m = _msg(fdp, "PreferredAllocationResponse")
    _field(m, "container_responses", 1, _F.TYPE_MESSAGE,
           _F.LABEL_REPEATED,
           ".v1beta1.ContainerPreferredAllocationResponse")

    _pool.Add(fdp)
    names = [
        "Empty", "DevicePluginOptions", "RegisterRequest", "NUMANode",
        "TopologyInfo", "Device", "ListAndWatchResponse",
        "ContainerAllocateRequest", "AllocateRequest", "Mount",
        "DeviceSpec", "ContainerAllocateResponse", "AllocateResponse",
        "PreStartContainerRequest", "PreStartContainerResponse",
        "ContainerPreferredAllocationRequest",
        "PreferredAllocationRequest",
        "ContainerPreferredAllocationResponse",
        "PreferredAllocationResponse",
    ]
    return {
        n: message_factory.GetMessageClass(
            _pool.FindMessageTypeByName(f"v1beta1.{n}"))
        for n in names
    }


_classes = _build()
globals().update(_classes)

DEVICE_PLUGIN_SERVICE = "v1beta1.DevicePlugin"
REGISTRATION_SERVICE = "v1beta1.Registration"
API_VERSION = "v1beta1"

__all__ = list(_classes) + [
    "DEVICE_PLUGIN_SERVICE", "REGISTRATION_SERVICE", "API_VERSION",
]
