"""kubelet device-plugin v1beta1 message definitions + gRPC plumbing.

Field numbers mirror k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1
api.proto exactly; the wire codec is util/pbwire.  Service wiring uses
grpcio generic handlers (no protoc needed).
"""
from __future__ import annotations

import grpc

from ..util.pbwire import (
    K_BOOL,
    K_INT,
    K_MAP_SS,
    K_MSG,
    K_STR,
    Message,
)

API_VERSION = "v1beta1"
KUBELET_SOCKET = "/var/lib/kubelet/device-plugins/kubelet.sock"
PLUGINS_DIR = "/var/lib/kubelet/device-plugins"

HEALTHY = "Healthy"
UNHEALTHY = "Unhealthy"


class Empty(Message):
    FIELDS = {}


class DevicePluginOptions(Message):
    FIELDS = {
        1: ("pre_start_required", K_BOOL, False, None),
        2: ("get_preferred_allocation_available", K_BOOL, False, None),
    }


class RegisterRequest(Message):
    FIELDS = {
        1: ("version", K_STR, False, None),
        2: ("endpoint", K_STR, False, None),
        3: ("resource_name", K_STR, False, None),
        4: ("options", K_MSG, False, DevicePluginOptions),
    }


class NUMANode(Message):
    FIELDS = {1: ("ID", K_INT, False, None)}


class TopologyInfo(Message):
    FIELDS = {1: ("nodes", K_MSG, True, NUMANode)}


class Device(Message):
    FIELDS = {
        1: ("ID", K_STR, False, None),
        2: ("health", K_STR, False, None),
        3: ("topology", K_MSG, False, TopologyInfo),
    }


class ListAndWatchResponse(Message):
    FIELDS = {1: ("devices", K_MSG, True, Device)}


class ContainerAllocateRequest(Message):
    FIELDS = {1: ("devices_ids", K_STR, True, None)}


class AllocateRequest(Message):
    FIELDS = {1: ("container_requests", K_MSG, True,
                  ContainerAllocateRequest)}


class Mount(Message):
    FIELDS = {
        1: ("container_path", K_STR, False, None),
        2: ("host_path", K_STR, False, None),
        3: ("read_only", K_BOOL, False, None),
    }


class DeviceSpec(Message):
    FIELDS = {
        1: ("container_path", K_STR, False, None),
        2: ("host_path", K_STR, False, None),
        3: ("permissions", K_STR, False, None),
    }


class CDIDevice(Message):
    FIELDS = {1: ("name", K_STR, False, None)}


class ContainerAllocateResponse(Message):
    FIELDS = {
        1: ("envs", K_MAP_SS, False, None),
        2: ("mounts", K_MSG, True, Mount),
        3: ("devices", K_MSG, True, DeviceSpec),
        4: ("annotations", K_MAP_SS, False, None),
        5: ("cdi_devices", K_MSG, True, CDIDevice),
    }


class AllocateResponse(Message):
    FIELDS = {1: ("container_responses", K_MSG, True,
                  ContainerAllocateResponse)}


class ContainerPreferredAllocationRequest(Message):
    FIELDS = {
        1: ("available_device_ids", K_STR, True, None),
        2: ("must_include_device_ids", K_STR, True, None),
        3: ("allocation_size", K_INT, False, None),
    }


class PreferredAllocationRequest(Message):
    FIELDS = {1: ("container_requests", K_MSG, True,
                  ContainerPreferredAllocationRequest)}


class ContainerPreferredAllocationResponse(Message):
    FIELDS = {1: ("device_ids", K_STR, True, None)}


class PreferredAllocationResponse(Message):
    FIELDS = {1: ("container_responses", K_MSG, True,
                  ContainerPreferredAllocationResponse)}


class PreStartContainerRequest(Message):
    FIELDS = {1: ("devices_ids", K_STR, True, None)}


class PreStartContainerResponse(Message):
    FIELDS = {}


def _m(cls_in, cls_out, fn, unary=True):
    if unary:
        return grpc.unary_unary_rpc_method_handler(
            fn, request_deserializer=cls_in.decode,
            response_serializer=lambda m: m.encode())
    return grpc.unary_stream_rpc_method_handler(
        fn, request_deserializer=cls_in.decode,
        response_serializer=lambda m: m.encode())


def device_plugin_handler(servicer) -> grpc.GenericRpcHandler:
    """servicer must implement: GetDevicePluginOptions, ListAndWatch
    (generator), GetPreferredAllocation, Allocate, PreStartContainer."""
    return grpc.method_handlers_generic_handler(
        "v1beta1.DevicePlugin",
        {
            "GetDevicePluginOptions":
                _m(Empty, DevicePluginOptions,
                   servicer.GetDevicePluginOptions),
            "ListAndWatch":
                _m(Empty, ListAndWatchResponse, servicer.ListAndWatch,
                   unary=False),
            "GetPreferredAllocation":
                _m(PreferredAllocationRequest,
                   PreferredAllocationResponse,
                   servicer.GetPreferredAllocation),
            "Allocate":
                _m(AllocateRequest, AllocateResponse, servicer.Allocate),
            "PreStartContainer":
                _m(PreStartContainerRequest, PreStartContainerResponse,
                   servicer.PreStartContainer),
        })


def register_with_kubelet(kubelet_socket: str, endpoint: str,
                          resource_name: str,
                          options: DevicePluginOptions) -> None:
    """One-shot Registration.Register call against kubelet."""
    req = RegisterRequest(version=API_VERSION, endpoint=endpoint,
                          resource_name=resource_name, options=options)
    channel = grpc.insecure_channel(f"unix://{kubelet_socket}")
    stub = channel.unary_unary(
        "/v1beta1.Registration/Register",
        request_serializer=lambda m: m.encode(),
        response_deserializer=Empty.decode)
    stub(req, timeout=10)
    channel.close()
