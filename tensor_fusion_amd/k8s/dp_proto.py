"""kubelet device-plugin v1beta1 protobuf messages, built at runtime.

grpcio-tools is not installed in this image, so the FileDescriptorProto
for k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto is constructed
programmatically (field numbers and wire types match the upstream proto,
which is what interop with a real kubelet requires — binary protobuf
doesn't carry field names). Message classes come from
google.protobuf.message_factory over a private DescriptorPool.

Exposed: MSG[name] message classes and the gRPC method path constants.
"""
from __future__ import annotations

from typing import Dict

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto

PKG = "v1beta1"
REGISTRATION_SERVICE = f"/{PKG}.Registration/Register"
DP = f"/{PKG}.DevicePlugin"
M_OPTIONS = f"{DP}/GetDevicePluginOptions"
M_LISTWATCH = f"{DP}/ListAndWatch"
M_ALLOCATE = f"{DP}/Allocate"
M_PREFERRED = f"{DP}/GetPreferredAllocation"
M_PRESTART = f"{DP}/PreStartContainer"

KUBELET_SOCKET = "/var/lib/kubelet/device-plugins/kubelet.sock"
PLUGIN_SOCKET_DIR = "/var/lib/kubelet/device-plugins"
API_VERSION = "v1beta1"


def _field(name: str, num: int, ftype: int, label: int = _F.LABEL_OPTIONAL,
           type_name: str = "") -> descriptor_pb2.FieldDescriptorProto:
    f = _F(name=name, number=num, type=ftype, label=label)
    if type_name:
        f.type_name = f".{PKG}.{type_name}"
    return f


def _msg(name: str, *fields) -> descriptor_pb2.DescriptorProto:
    m = descriptor_pb2.DescriptorProto(name=name)
    m.field.extend(fields)
    return m


def _map_entry(parent: descriptor_pb2.DescriptorProto, field_name: str,
               num: int):
    """Add a map<string,string> field + its nested MapEntry message."""

    entry_name = "".join(p.title() for p in field_name.split("_")) + "Entry"
    entry = descriptor_pb2.DescriptorProto(name=entry_name)
    entry.field.extend([
        _field("key", 1, _F.TYPE_STRING),
        _field("value", 2, _F.TYPE_STRING),
    ])
    entry.options.map_entry = True
    parent.nested_type.append(entry)
    f = _F(name=field_name, number=num, type=_F.TYPE_MESSAGE,
           label=_F.LABEL_REPEATED)
    f.type_name = f".{PKG}.{parent.name}.{entry_name}"
    parent.field.append(f)


def _build_pool():
    fd = descriptor_pb2.FileDescriptorProto(
        name="tensorfusion/deviceplugin_v1beta1.proto", package=PKG,
        syntax="proto3")

    fd.message_type.append(_msg("Empty"))

    fd.message_type.append(_msg(
        "DevicePluginOptions",
        _field("pre_start_required", 1, _F.TYPE_BOOL),
        _field("get_preferred_allocation_available", 2, _F.TYPE_BOOL),
    ))

    fd.message_type.append(_msg(
        "RegisterRequest",
        _field("version", 1, _F.TYPE_STRING),
        _field("endpoint", 2, _F.TYPE_STRING),
        _field("resource_name", 3, _F.TYPE_STRING),
        _field("options", 4, _F.TYPE_MESSAGE,
               type_name="DevicePluginOptions"),
    ))

    fd.message_type.append(_msg(
        "NUMANode", _field("ID", 1, _F.TYPE_INT64)))
    fd.message_type.append(_msg(
        "TopologyInfo",
        _field("nodes", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="NUMANode")))
    fd.message_type.append(_msg(
        "Device",
        _field("ID", 1, _F.TYPE_STRING),
        _field("health", 2, _F.TYPE_STRING),
        _field("topology", 3, _F.TYPE_MESSAGE, type_name="TopologyInfo"),
    ))
    fd.message_type.append(_msg(
        "ListAndWatchResponse",
        _field("devices", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="Device")))

    fd.message_type.append(_msg(
        "ContainerAllocateRequest",
        _field("devicesIDs", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)))
    fd.message_type.append(_msg(
        "AllocateRequest",
        _field("container_requests", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="ContainerAllocateRequest")))

    fd.message_type.append(_msg(
        "Mount",
        _field("container_path", 1, _F.TYPE_STRING),
        _field("host_path", 2, _F.TYPE_STRING),
        _field("read_only", 3, _F.TYPE_BOOL),
    ))
    fd.message_type.append(_msg(
        "DeviceSpec",
        _field("container_path", 1, _F.TYPE_STRING),
        _field("host_path", 2, _F.TYPE_STRING),
        _field("permissions", 3, _F.TYPE_STRING),
    ))
    fd.message_type.append(_msg(
        "CDIDevice", _field("name", 1, _F.TYPE_STRING)))

    car = _msg(
        "ContainerAllocateResponse")
    _map_entry(car, "envs", 1)
    car.field.append(_field("mounts", 2, _F.TYPE_MESSAGE,
                            _F.LABEL_REPEATED, type_name="Mount"))
    car.field.append(_field("devices", 3, _F.TYPE_MESSAGE,
                            _F.LABEL_REPEATED, type_name="DeviceSpec"))
    _map_entry(car, "annotations", 4)
    car.field.append(_field("cdi_devices", 5, _F.TYPE_MESSAGE,
                            _F.LABEL_REPEATED, type_name="CDIDevice"))
    fd.message_type.append(car)

    fd.message_type.append(_msg(
        "AllocateResponse",
        _field("container_responses", 1, _F.TYPE_MESSAGE,
               _F.LABEL_REPEATED, type_name="ContainerAllocateResponse")))

    fd.message_type.append(_msg(
        "PreStartContainerRequest",
        _field("devicesIDs", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)))
    fd.message_type.append(_msg("PreStartContainerResponse"))

    fd.message_type.append(_msg(
        "ContainerPreferredAllocationRequest",
        _field("available_deviceIDs", 1, _F.TYPE_STRING, _F.LABEL_REPEATED),
        _field("must_include_deviceIDs", 2, _F.TYPE_STRING,
               _F.LABEL_REPEATED),
        _field("allocation_size", 3, _F.TYPE_INT32),
    ))
    fd.message_type.append(_msg(
        "PreferredAllocationRequest",
        _field("container_requests", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="ContainerPreferredAllocationRequest")))
    fd.message_type.append(_msg(
        "ContainerPreferredAllocationResponse",
        _field("deviceIDs", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)))
    fd.message_type.append(_msg(
        "PreferredAllocationResponse",
        _field("container_responses", 1, _F.TYPE_MESSAGE,
               _F.LABEL_REPEATED,
               type_name="ContainerPreferredAllocationResponse")))

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fd)
    names = [m.name for m in fd.message_type]
    msgs: Dict[str, type] = {}
    for n in names:
        desc = pool.FindMessageTypeByName(f"{PKG}.{n}")
        msgs[n] = message_factory.GetMessageClass(desc)
    return msgs


MSG: Dict[str, type] = _build_pool()
