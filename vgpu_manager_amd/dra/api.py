"""DRA kubelet gRPC API (k8s.io/kubelet/pkg/apis/dra/v1beta1) +
kubelet pluginregistration v1 — wire-compatible via util/pbwire
(reference cmd/kubelet-plugin serves the same two services through
k8s.io/dynamic-resource-allocation/kubeletplugin).

Services:
  v1beta1.DRAPlugin/NodePrepareResources, NodeUnprepareResources
  pluginregistration.Registration/GetInfo, NotifyRegistrationStatus
"""
from __future__ import annotations

import grpc

from ..util.pbwire import K_BOOL, K_MSG, K_STR, Message

DRA_SERVICE = "v1beta1.DRAPlugin"
REGISTRATION_SERVICE = "pluginregistration.Registration"
PLUGIN_TYPE_DRA = "DRAPlugin"
PLUGINS_REGISTRY = "/var/lib/kubelet/plugins_registry"
PLUGINS_DIR = "/var/lib/kubelet/plugins"


class Claim(Message):
    FIELDS = {
        1: ("namespace", K_STR, False, None),
        2: ("uid", K_STR, False, None),
        3: ("name", K_STR, False, None),
    }


class NodePrepareResourcesRequest(Message):
    FIELDS = {1: ("claims", K_MSG, True, Claim)}


class Device(Message):
    FIELDS = {
        1: ("request_names", K_STR, True, None),
        2: ("pool_name", K_STR, False, None),
        3: ("device_name", K_STR, False, None),
        4: ("cdi_device_ids", K_STR, True, None),
    }


class NodePrepareResourceResponse(Message):
    FIELDS = {
        1: ("devices", K_MSG, True, Device),
        2: ("error", K_STR, False, None),
    }


class _PrepareEntry(Message):
    """map<string, NodePrepareResourceResponse> entry."""
    FIELDS = {
        1: ("key", K_STR, False, None),
        2: ("value", K_MSG, False, NodePrepareResourceResponse),
    }


class NodePrepareResourcesResponse(Message):
    FIELDS = {1: ("claims", K_MSG, True, _PrepareEntry)}


class NodeUnprepareResourcesRequest(Message):
    FIELDS = {1: ("claims", K_MSG, True, Claim)}


class NodeUnprepareResourceResponse(Message):
    FIELDS = {1: ("error", K_STR, False, None)}


class _UnprepareEntry(Message):
    FIELDS = {
        1: ("key", K_STR, False, None),
        2: ("value", K_MSG, False, NodeUnprepareResourceResponse),
    }


class NodeUnprepareResourcesResponse(Message):
    FIELDS = {1: ("claims", K_MSG, True, _UnprepareEntry)}


# ---- pluginregistration v1 ----

class InfoRequest(Message):
    FIELDS = {}


class PluginInfo(Message):
    FIELDS = {
        1: ("type", K_STR, False, None),
        2: ("name", K_STR, False, None),
        3: ("endpoint", K_STR, False, None),
        4: ("supported_versions", K_STR, True, None),
    }


class RegistrationStatus(Message):
    FIELDS = {
        1: ("plugin_registered", K_BOOL, False, None),
        2: ("error", K_STR, False, None),
    }


class RegistrationStatusResponse(Message):
    FIELDS = {}


def _u(cls_in, fn):
    return grpc.unary_unary_rpc_method_handler(
        fn, request_deserializer=cls_in.decode,
        response_serializer=lambda m: m.encode())


def dra_plugin_handler(servicer) -> grpc.GenericRpcHandler:
    """servicer: NodePrepareResources, NodeUnprepareResources."""
    return grpc.method_handlers_generic_handler(
        DRA_SERVICE,
        {
            "NodePrepareResources":
                _u(NodePrepareResourcesRequest,
                   servicer.NodePrepareResources),
            "NodeUnprepareResources":
                _u(NodeUnprepareResourcesRequest,
                   servicer.NodeUnprepareResources),
        })


def registration_handler(servicer) -> grpc.GenericRpcHandler:
    """servicer: GetInfo, NotifyRegistrationStatus (kubelet's plugin
    watcher dials the registry socket and calls these)."""
    return grpc.method_handlers_generic_handler(
        REGISTRATION_SERVICE,
        {
            "GetInfo": _u(InfoRequest, servicer.GetInfo),
            "NotifyRegistrationStatus":
                _u(RegistrationStatus,
                   servicer.NotifyRegistrationStatus),
        })
