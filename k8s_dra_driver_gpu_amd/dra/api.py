"""DRA kubelet-plugin gRPC message types + service plumbing.

Wire-compatible with ``k8s.io/kubelet/pkg/apis/dra/v1beta1``: the gRPC
service is registered under its fully-qualified proto name
``k8s.io.kubelet.pkg.apis.dra.v1beta1.DRAPlugin`` (and the v1 surface under
``k8s.io.kubelet.pkg.apis.dra.v1.DRAPlugin``), exactly as the kubelet dials
it (ref vendor k8s.io/kubelet/pkg/apis/dra/v1beta1/api.proto:21 ``package
k8s.io.kubelet.pkg.apis.dra.v1beta1``; api_grpc.pb.go:55-56,190
``ServiceName: "k8s.io.kubelet.pkg.apis.dra.v1beta1.DRAPlugin"``).

Plugin registration speaks ``pluginregistration/v1`` (service
``pluginregistration.Registration``) and advertises the DRA *service
identifiers* ``v1.DRAPlugin`` / ``v1beta1.DRAPlugin`` as its
"supported versions" — NOT bare API versions (ref vendor
k8s.io/kubelet/pkg/apis/dra/v1beta1/types.go:23 ``DRAPluginService =
"v1beta1.DRAPlugin"``; dynamic-resource-allocation/kubeletplugin/
draplugin.go:755-763 appends v1 then v1beta1).
"""

from __future__ import annotations

from typing import Callable, Optional

import grpc

from .protowire import Message

# ---------------------------------------------------------------------------
# DRA v1beta1 messages
# ---------------------------------------------------------------------------


class Claim(Message):
    FIELDS = {1: ("namespace", "string"), 2: ("uid", "string"), 3: ("name", "string")}


class NodePrepareResourcesRequest(Message):
    FIELDS = {1: ("claims", ("repeated", Claim))}


class Device(Message):
    FIELDS = {
        1: ("request_names", ("repeated", "string")),
        2: ("pool_name", "string"),
        3: ("device_name", "string"),
        4: ("cdi_device_ids", ("repeated", "string")),
    }


class NodePrepareResourceResponse(Message):
    FIELDS = {1: ("devices", ("repeated", Device)), 2: ("error", "string")}


class NodePrepareResourcesResponse(Message):
    FIELDS = {1: ("claims", ("map", "string", NodePrepareResourceResponse))}


class NodeUnprepareResourcesRequest(Message):
    FIELDS = {1: ("claims", ("repeated", Claim))}


class NodeUnprepareResourceResponse(Message):
    FIELDS = {1: ("error", "string")}


class NodeUnprepareResourcesResponse(Message):
    FIELDS = {1: ("claims", ("map", "string", NodeUnprepareResourceResponse))}


# ---------------------------------------------------------------------------
# Plugin registration messages
# ---------------------------------------------------------------------------


class InfoRequest(Message):
    FIELDS = {}


class PluginInfo(Message):
    FIELDS = {
        1: ("type", "string"),
        2: ("name", "string"),
        3: ("endpoint", "string"),
        4: ("supported_versions", ("repeated", "string")),
    }


class RegistrationStatus(Message):
    FIELDS = {1: ("plugin_registered", "bool"), 2: ("error", "string")}


class RegistrationStatusResponse(Message):
    FIELDS = {}


DRA_PLUGIN_TYPE = "DRAPlugin"  # kubelet plugin registration type

# Fully-qualified gRPC service names (proto package + service). The kubelet's
# client stubs invoke e.g.
# /k8s.io.kubelet.pkg.apis.dra.v1beta1.DRAPlugin/NodePrepareResources
# (ref vendor .../dra/v1beta1/api_grpc.pb.go:55-56,190; v1: api_grpc.pb.go:190).
DRA_SERVICE_FULL = {
    "v1": "k8s.io.kubelet.pkg.apis.dra.v1.DRAPlugin",
    "v1beta1": "k8s.io.kubelet.pkg.apis.dra.v1beta1.DRAPlugin",
}

# Service *identifiers* advertised through pluginregistration GetInfo's
# supported_versions field — the kubelet matches these strings to decide which
# DRA service variant to dial (ref vendor .../dra/{v1,v1beta1}/types.go:23,
# kubeletplugin/noderegistrar.go:39). Order matches the reference helper:
# v1 first, then v1beta1 (draplugin.go:755-763).
DRA_SERVICE_IDENTIFIERS = ("v1.DRAPlugin", "v1beta1.DRAPlugin")


def _unary(handler: Callable[[Message], Message], req_cls):
    def call(request_bytes, context):
        req = req_cls.from_bytes(request_bytes)
        return handler(req, context).to_bytes()

    return grpc.unary_unary_rpc_method_handler(
        call, request_deserializer=lambda b: b, response_serializer=lambda b: b
    )


class DRAPluginServicer:
    """Implement node_prepare_resources / node_unprepare_resources."""

    def node_prepare_resources(
        self, req: NodePrepareResourcesRequest, context
    ) -> NodePrepareResourcesResponse:
        raise NotImplementedError

    def node_unprepare_resources(
        self, req: NodeUnprepareResourcesRequest, context
    ) -> NodeUnprepareResourcesResponse:
        raise NotImplementedError

    def add_to_server(self, server: grpc.Server, packages=("v1beta1", "v1")) -> None:
        handlers = {
            "NodePrepareResources": _unary(self.node_prepare_resources, NodePrepareResourcesRequest),
            "NodeUnprepareResources": _unary(
                self.node_unprepare_resources, NodeUnprepareResourcesRequest
            ),
        }
        for pkg in packages:
            server.add_generic_rpc_handlers(
                (grpc.method_handlers_generic_handler(DRA_SERVICE_FULL[pkg], handlers),)
            )


class RegistrationServicer:
    """kubelet pluginregistration/v1 service (ref: kubeletplugin helper
    registers via the plugins_registry socket)."""

    def __init__(self, name: str, endpoint: str, supported_versions=DRA_SERVICE_IDENTIFIERS):
        self.name = name
        self.endpoint = endpoint
        self.supported_versions = list(supported_versions)
        self.registered: Optional[bool] = None
        self.register_error: str = ""

    def get_info(self, req: InfoRequest, context) -> PluginInfo:
        return PluginInfo(
            type=DRA_PLUGIN_TYPE,
            name=self.name,
            endpoint=self.endpoint,
            supported_versions=self.supported_versions,
        )

    def notify_registration_status(self, req: RegistrationStatus, context):
        self.registered = req.plugin_registered
        self.register_error = req.error
        return RegistrationStatusResponse()

    def add_to_server(self, server: grpc.Server) -> None:
        handlers = {
            "GetInfo": _unary(self.get_info, InfoRequest),
            "NotifyRegistrationStatus": _unary(
                self.notify_registration_status, RegistrationStatus
            ),
        }
        server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler("pluginregistration.Registration", handlers),)
        )


class DRAPluginClient:
    """Raw-bytes gRPC client for the DRA service (used by tests, the bench
    harness's fake kubelet, and the healthcheck self-probe)."""

    def __init__(self, target: str, package: str = "v1beta1", channel: Optional[grpc.Channel] = None):
        self.channel = channel or grpc.insecure_channel(target)
        p = f"/{DRA_SERVICE_FULL[package]}"
        self._prepare = self.channel.unary_unary(
            f"{p}/NodePrepareResources",
            request_serializer=lambda m: m.to_bytes(),
            response_deserializer=NodePrepareResourcesResponse.from_bytes,
        )
        self._unprepare = self.channel.unary_unary(
            f"{p}/NodeUnprepareResources",
            request_serializer=lambda m: m.to_bytes(),
            response_deserializer=NodeUnprepareResourcesResponse.from_bytes,
        )

    def prepare(self, claims, timeout: float = 60.0) -> NodePrepareResourcesResponse:
        return self._prepare(NodePrepareResourcesRequest(claims=claims), timeout=timeout)

    def unprepare(self, claims, timeout: float = 60.0) -> NodeUnprepareResourcesResponse:
        return self._unprepare(NodeUnprepareResourcesRequest(claims=claims), timeout=timeout)

    def close(self):
        self.channel.close()


class RegistrationClient:
    def __init__(self, target: str, channel: Optional[grpc.Channel] = None):
        self.channel = channel or grpc.insecure_channel(target)
        self._get_info = self.channel.unary_unary(
            "/pluginregistration.Registration/GetInfo",
            request_serializer=lambda m: m.to_bytes(),
            response_deserializer=PluginInfo.from_bytes,
        )
        self._notify = self.channel.unary_unary(
            "/pluginregistration.Registration/NotifyRegistrationStatus",
            request_serializer=lambda m: m.to_bytes(),
            response_deserializer=RegistrationStatusResponse.from_bytes,
        )

    def get_info(self, timeout: float = 10.0) -> PluginInfo:
        return self._get_info(InfoRequest(), timeout=timeout)

    def notify(self, registered: bool, error: str = "", timeout: float = 10.0):
        return self._notify(
            RegistrationStatus(plugin_registered=registered, error=error), timeout=timeout
        )

    def close(self):
        self.channel.close()
