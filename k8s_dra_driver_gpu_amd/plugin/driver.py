"""GPU kubelet-plugin driver: gRPC glue between the kubelet DRA contract and
the device state machine.

Parity with the reference's ``cmd/gpu-kubelet-plugin/driver.go``:
``NewDriver`` wiring (:70-186), per-claim prepare under the node-global
prepare/unprepare lock (:373-418, the flock lives in DeviceState), per-claim
error isolation in batch responses (:337-371), registration with the kubelet
via the plugins-registry socket, and ResourceSlice publication
(:455-494; the slice model is in ``resourceslice.py``, publication goes
through the k8s client when attached).
"""

from __future__ import annotations

import logging
import os
import time
from concurrent import futures
from typing import Callable, Dict, List, Optional

import grpc

from .. import GPU_DRIVER_NAME
from ..dra import api as dra
from ..metrics.dra import DraMetrics
from .checkpoint import ClaimRef
from .device_state import AllocatedClaim, AllocatedDevice, DeviceState

logger = logging.getLogger("amddra.driver")

# resolver: (namespace, name, uid) -> AllocatedClaim (reads the ResourceClaim's
# allocation result from the API server in production; fakes in tests/bench)
ClaimResolver = Callable[[str, str, str], AllocatedClaim]


class GpuDriver(dra.DRAPluginServicer):
    def __init__(
        self,
        state: DeviceState,
        claim_resolver: ClaimResolver,
        node_name: str = "",
        driver_name: str = GPU_DRIVER_NAME,
        metrics: Optional[DraMetrics] = None,
    ):
        self.state = state
        self.claim_resolver = claim_resolver
        self.node_name = node_name or os.environ.get("NODE_NAME", "node")
        self.driver_name = driver_name
        self.metrics = metrics or DraMetrics()
        self._server: Optional[grpc.Server] = None
        self._reg_server: Optional[grpc.Server] = None
        self.registration: Optional[dra.RegistrationServicer] = None

    # -- DRA service -------------------------------------------------------

    def node_prepare_resources(self, req, context):
        resp = dra.NodePrepareResourcesResponse()
        for claim in req.claims:
            self.metrics.requests_inflight.inc()
            t0 = time.monotonic()
            try:
                allocated = self.claim_resolver(claim.namespace, claim.name, claim.uid)
                results = self.state.prepare(allocated)
                devices = [
                    dra.Device(
                        request_names=[r.request] if r.request else [],
                        pool_name=self.node_name,
                        device_name=r.device,
                        cdi_device_ids=r.cdi_device_ids,
                    )
                    for r in results
                ]
                resp.claims[claim.uid] = dra.NodePrepareResourceResponse(devices=devices)
                self.metrics.requests_total.labels("prepare", "success").inc()
                self._update_prepared_gauge()
            except Exception as e:
                logger.exception("prepare failed for claim %s/%s", claim.namespace, claim.name)
                resp.claims[claim.uid] = dra.NodePrepareResourceResponse(error=str(e))
                self.metrics.requests_total.labels("prepare", "error").inc()
                self.metrics.prepare_errors_total.inc()
            finally:
                self.metrics.request_duration.labels("prepare").observe(time.monotonic() - t0)
                self.metrics.requests_inflight.dec()
        return resp

    def node_unprepare_resources(self, req, context):
        resp = dra.NodeUnprepareResourcesResponse()
        for claim in req.claims:
            self.metrics.requests_inflight.inc()
            t0 = time.monotonic()
            try:
                self.state.unprepare(claim.uid)
                resp.claims[claim.uid] = dra.NodeUnprepareResourceResponse()
                self.metrics.requests_total.labels("unprepare", "success").inc()
                self._update_prepared_gauge()
            except Exception as e:
                logger.exception("unprepare failed for claim %s", claim.uid)
                resp.claims[claim.uid] = dra.NodeUnprepareResourceResponse(error=str(e))
                self.metrics.requests_total.labels("unprepare", "error").inc()
                self.metrics.unprepare_errors_total.inc()
            finally:
                self.metrics.request_duration.labels("unprepare").observe(time.monotonic() - t0)
                self.metrics.requests_inflight.dec()
        return resp

    def _update_prepared_gauge(self) -> None:
        """prepared_devices gauge by type (ref pkg/metrics prepared_devices)."""
        try:
            counts: Dict[str, int] = {}
            for pc in self.state.prepared_claims().values():
                for d in pc.devices or []:
                    counts[d.type] = counts.get(d.type, 0) + 1
            for t in ("gpu", "partition", "vfio"):
                self.metrics.prepared_devices.labels(type=t).set(counts.get(t, 0))
        except Exception:
            logger.debug("prepared-devices gauge update failed", exc_info=True)

    # -- serving ------------------------------------------------------------

    def start(
        self,
        plugin_dir: str,
        registry_dir: str = "",
        workers: int = 4,
    ) -> Dict[str, str]:
        """Serve the DRA socket in `plugin_dir` and (optionally) the
        registration socket in `registry_dir`; returns the socket paths."""
        os.makedirs(plugin_dir, exist_ok=True)
        dra_sock = os.path.join(plugin_dir, "dra.sock")
        self._remove_stale(dra_sock)
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=workers))
        self.add_to_server(self._server)
        from ..utils.paths import check_unix_socket_path
        self._server.add_insecure_port(f"unix://{check_unix_socket_path(dra_sock)}")
        self._server.start()
        out = {"dra": dra_sock}

        if registry_dir:
            os.makedirs(registry_dir, exist_ok=True)
            reg_sock = os.path.join(registry_dir, f"{self.driver_name}-reg.sock")
            self._remove_stale(reg_sock)
            self.registration = dra.RegistrationServicer(
                name=self.driver_name, endpoint=dra_sock
            )
            self._reg_server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
            self.registration.add_to_server(self._reg_server)
            self._reg_server.add_insecure_port(f"unix://{check_unix_socket_path(reg_sock)}")
            self._reg_server.start()
            out["registration"] = reg_sock
        return out

    @staticmethod
    def _remove_stale(path: str) -> None:
        try:
            os.unlink(path)
        except FileNotFoundError:
            pass

    def stop(self, grace: float = 1.0) -> None:
        if self._server:
            self._server.stop(grace)
        if self._reg_server:
            self._reg_server.stop(grace)


def k8s_claim_resolver(client, driver_name: str = GPU_DRIVER_NAME) -> ClaimResolver:
    """Production resolver: read the ResourceClaim's allocation result from
    the API server and convert it to an AllocatedClaim (the reference reads
    the same data through its informers; device_state.go:697
    GetOpaqueDeviceConfigs consumes status.allocation.devices)."""

    def resolve(namespace: str, name: str, uid: str) -> AllocatedClaim:
        obj = client.get_or_none("resourceclaims", name, namespace)
        if obj is None:
            raise KeyError(f"resourceclaim {namespace}/{name} not found")
        if uid and obj.get("metadata", {}).get("uid") not in ("", uid):
            raise KeyError(f"resourceclaim {namespace}/{name} uid mismatch")
        alloc = (obj.get("status") or {}).get("allocation") or {}
        results = ((alloc.get("devices") or {}).get("results")) or []
        configs = ((alloc.get("devices") or {}).get("config")) or []
        devices = []
        for res in results:
            if res.get("driver") not in (None, "", driver_name):
                continue
            request = res.get("request", "")
            dev_configs = []
            for c in configs:
                opaque = c.get("opaque") or {}
                if opaque.get("driver") != driver_name:
                    continue
                reqs = c.get("requests") or []
                if not reqs or request in reqs:
                    dev_configs.append(opaque.get("parameters"))
            devices.append(
                AllocatedDevice(device=res.get("device", ""), configs=dev_configs, request=request)
            )
        if not devices:
            raise KeyError(
                f"resourceclaim {namespace}/{name} has no allocation for driver {driver_name}"
            )
        return AllocatedClaim(
            ref=ClaimRef(namespace=namespace, name=name, uid=uid or obj["metadata"].get("uid", "")),
            devices=devices,
        )

    return resolve


def static_claim_resolver(store: Dict[str, AllocatedClaim]) -> ClaimResolver:
    """Test/bench resolver backed by a dict keyed on claim UID."""

    def resolve(namespace: str, name: str, uid: str) -> AllocatedClaim:
        claim = store.get(uid)
        if claim is None:
            raise KeyError(f"no allocation recorded for claim {namespace}/{name} uid={uid}")
        return claim

    return resolve
