"""LocalCluster: the whole driver stack in one process against the fake API
server — the analog of the reference's kind + mock-NVML CI cluster
(``hack/ci/mock-nvml``, ``demo/clusters``).

Simulates the pieces a real cluster provides (scheduler via the stub,
kubelet claim handling, DaemonSet pod launching) while running OUR components
unmodified: controller, both kubelet plugins over real gRPC sockets, daemon
supervisors with the real C++ fabricd, CDI, checkpoints and the mock (or
real) device layer. Used by ``demo/run_local.py`` and integration tests.
"""

from __future__ import annotations

import logging
import os
import socket
import tempfile
import threading
import time
from typing import Any, Dict, List, Optional

import yaml

from ..cdi.spec import CdiHandler
from ..cdplugin.plugin import ComputeDomainPlugin
from ..controller.computedomain import ComputeDomainController
from ..daemon.main import DaemonSupervisor
from ..daemon.process import default_fabricd_path
from ..device.devicelib import DeviceLib
from ..device.mock import MockTree
from ..dra import api as dra
from ..k8s.client import FakeClient
from ..k8s.scheduler import SchedulerStub
from ..plugin.checkpoint import CheckpointManager
from ..plugin.device_state import DeviceState
from ..plugin.driver import GpuDriver, k8s_claim_resolver
from ..plugin.resourceslice import ResourceSliceGenerator
from ..webhook.server import validate_admission_review

logger = logging.getLogger("amddra.localcluster")

CHART_DIR = os.path.join(
    os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))),
    "deployments", "helm", "amd-dra-driver",
)

KIND_TO_RESOURCE = {
    "Namespace": None,  # implicit
    "ComputeDomain": "computedomains",
    "ComputeDomainClique": "computedomaincliques",
    "ResourceClaim": "resourceclaims",
    "ResourceClaimTemplate": "resourceclaimtemplates",
    "ResourceSlice": "resourceslices",
    "DeviceClass": "deviceclasses",
    "DaemonSet": "daemonsets",
    "Deployment": "deployments",
    "Pod": "pods",
    "Node": "nodes",
    "Job": "jobs",
}


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


class LocalCluster:
    def __init__(self, num_gpus: int = 8, node_name: str = "node-a",
                 work_dir: str = "", real_devices: bool = False,
                 partitionable: bool = True, vfio: bool = False):
        self.client = FakeClient()
        self.node_name = node_name
        self.work_dir = work_dir or tempfile.mkdtemp(prefix="amddra-local-")
        self.mock: Optional[MockTree] = None
        if real_devices:
            self.devicelib = DeviceLib()
        else:
            self.mock = MockTree(root=os.path.join(self.work_dir, "mock"), num_gpus=num_gpus)
            self.mock.setup()
            self.devicelib = DeviceLib(backend=self.mock.backend())
        self.partitionable = partitionable
        self.vfio = vfio
        self.scheduler = SchedulerStub(self.client)
        self.controller: Optional[ComputeDomainController] = None
        self.gpu_driver: Optional[GpuDriver] = None
        self.gpu_client: Optional[dra.DRAPluginClient] = None
        self.cd_plugin: Optional[ComputeDomainPlugin] = None
        self.supervisors: Dict[str, DaemonSupervisor] = {}
        self._threads: List[threading.Thread] = []
        self._prepared_pods: Dict[str, List[str]] = {}
        self._pod_claims: Dict[str, List[str]] = {}
        # how long _run_workload polls the scheduler before declaring a claim
        # unschedulable (tests shrink this)
        self.schedule_timeout = 30.0

    # -- lifecycle ----------------------------------------------------------

    def start(self) -> "LocalCluster":
        # DeviceClasses + node object (chart rendered via helmlite; this
        # cluster models k8s >= 1.35, so classes serve under resource.k8s.io/v1
        # with extendedResourceName on gpu.amd.com)
        from ..utils.helmlite import render_chart

        rendered = render_chart(CHART_DIR, {"resourceApiVersion": "v1"})
        for doc in yaml.safe_load_all(rendered["deviceclasses.yaml"]):
            if doc:
                self.client.create("deviceclasses", doc)
        self.client.create(
            "nodes", {"apiVersion": "v1", "kind": "Node",
                      "metadata": {"name": self.node_name}}
        )
        # controller
        self.controller = ComputeDomainController(
            self.client, status_sync_period=0.2, cleanup_period=3600
        ).start()
        # GPU plugin
        state_dir = os.path.join(self.work_dir, "gpu-plugin")
        dev_root = self.mock.dev_root if self.mock else "/dev"
        ds = DeviceState(
            devicelib=self.devicelib,
            cdi=CdiHandler(cdi_root=os.path.join(self.work_dir, "cdi"), dev_root=dev_root),
            checkpoints=CheckpointManager(state_dir),
            state_dir=state_dir,
            vfio=(self.mock.vfio_manager() if (self.vfio and self.mock) else None),
        )
        self.gpu_driver = GpuDriver(
            state=ds, claim_resolver=k8s_claim_resolver(self.client),
            node_name=self.node_name,
        )
        socks = self.gpu_driver.start(plugin_dir=os.path.join(self.work_dir, "plugin"))
        self.gpu_client = dra.DRAPluginClient(f"unix://{socks['dra']}")
        for sl in ResourceSliceGenerator(
            self.devicelib, node_name=self.node_name, partitionable=self.partitionable,
            vfio=self.vfio
        ).generate():
            self.client.apply("resourceslices", sl)
        # CD plugin
        self.cd_plugin = ComputeDomainPlugin(
            client=self.client, devicelib=self.devicelib,
            state_dir=os.path.join(self.work_dir, "cd-plugin"),
            node_name=self.node_name, retry_max_timeout=10.0,
        )
        self.client.apply("resourceslices", self.cd_plugin.resource_slice())
        # "kubelet" for DaemonSets: launch a daemon supervisor per matching DS
        t = threading.Thread(target=self._daemonset_kubelet, daemon=True)
        t.start()
        self._threads.append(t)
        return self

    def stop(self) -> None:
        for sup in self.supervisors.values():
            sup.stop()
        if self.controller:
            self.controller.stop()
        if self.gpu_client:
            self.gpu_client.close()
        if self.gpu_driver:
            self.gpu_driver.stop()

    # -- DaemonSet simulation -------------------------------------------------

    def _daemonset_kubelet(self) -> None:
        """Launch a DaemonSupervisor for each compute-domain DaemonSet whose
        node selector matches (after 'scheduling' the node into the CD —
        in a real cluster the workload channel prepare labels the node; for
        daemon placement the controller's DS nodeSelector requires the label,
        which we set here as the scheduler would on daemon placement)."""
        from ..controller.templates import CD_LABEL_KEY

        while True:
            time.sleep(0.05)  # kubelet reacts to DS pod creation near-instantly
            try:
                for ds in self.client.list("daemonsets"):
                    labels = ds["metadata"].get("labels") or {}
                    cd_uid = labels.get(CD_LABEL_KEY)
                    if not cd_uid or cd_uid in self.supervisors:
                        continue
                    self.client.patch(
                        "nodes", self.node_name,
                        {"metadata": {"labels": {CD_LABEL_KEY: cd_uid}}},
                    )
                    sup = DaemonSupervisor(
                        client=self.client, cd_uid=cd_uid, node_name=self.node_name,
                        pod_ip="127.0.0.1",
                        work_dir=os.path.join(self.work_dir, f"fabricd-{cd_uid[:8]}"),
                        devicelib=self.devicelib,
                        peer_port=_free_port(), command_port=_free_port(),
                        fabricd_path=default_fabricd_path(),
                        gpu_probe=self.mock is None,  # real devices: probe-gated readiness
                    )
                    self.supervisors[cd_uid] = sup
                    t = threading.Thread(
                        target=lambda s=sup: s.run(ready_poll_interval=0.3), daemon=True
                    )
                    t.start()
                    self._threads.append(t)
                live = {cd["metadata"]["uid"] for cd in self.client.list("computedomains")}
                for uid in list(self.supervisors):
                    if uid not in live:
                        self.supervisors.pop(uid).stop()
            except Exception:
                logger.exception("daemonset kubelet loop error")

    # -- YAML application ------------------------------------------------------

    def apply_yaml(self, path: str) -> List[str]:
        """Apply a demo spec; returns human-readable event lines."""
        events = []
        with open(path) as f:
            docs = [d for d in yaml.safe_load_all(f) if d]
        for doc in docs:
            kind = doc.get("kind", "")
            if kind == "Namespace":
                continue
            # webhook admission for claims/templates
            if kind in ("ResourceClaim", "ResourceClaimTemplate"):
                review = {
                    "request": {
                        "uid": "local",
                        "kind": {"group": "resource.k8s.io", "version": "v1beta1",
                                 "kind": kind},
                        "object": doc,
                    }
                }
                out = validate_admission_review(review)
                if not out["response"]["allowed"]:
                    events.append(f"DENIED {kind} {doc['metadata'].get('name')}: "
                                  f"{out['response']['status']['message']}")
                    continue
            if kind in ("Pod", "Deployment", "Job"):
                events.extend(self._run_workload(doc))
                continue
            resource = KIND_TO_RESOURCE.get(kind)
            if resource is None:
                events.append(f"SKIP {kind}")
                continue
            self.client.apply(resource, doc)
            events.append(f"APPLIED {kind} {doc['metadata'].get('name')}")
        return events

    # -- Pod simulation ---------------------------------------------------------

    def _pod_specs(self, doc) -> List[Dict[str, Any]]:
        kind = doc.get("kind")
        ns = doc["metadata"].get("namespace", "default")
        if kind == "Pod":
            return [dict(doc, metadata={**doc["metadata"], "namespace": ns})]
        template = doc["spec"]["template"]
        replicas = doc["spec"].get("replicas", 1) if kind == "Deployment" else 1
        pods = []
        for i in range(replicas):
            pods.append(
                {
                    "apiVersion": "v1",
                    "kind": "Pod",
                    "metadata": {
                        "name": f"{doc['metadata']['name']}-{i}",
                        "namespace": ns,
                        "labels": (template.get("metadata") or {}).get("labels", {}),
                    },
                    "spec": template["spec"],
                }
            )
        return pods

    def _run_workload(self, doc) -> List[str]:
        """Simulate kubelet handling: create claims from templates, schedule,
        NodePrepareResources over real gRPC, record CDI ids."""
        events = []
        for pod in self._pod_specs(doc):
            ns = pod["metadata"]["namespace"]
            pod_name = pod["metadata"]["name"]
            self.client.apply("pods", pod)
            cdi_ids = []
            for rc in pod["spec"].get("resourceClaims") or []:
                if rc.get("resourceClaimName"):
                    # shared claim: N pods reference one pre-created claim
                    # (gpu-test2/e2e shared-claim scenario)
                    claim_name = rc["resourceClaimName"]
                    self.scheduler.schedule_pending()
                    claim = self.client.get_or_none("resourceclaims", claim_name, ns)
                    if claim is None or not (claim.get("status") or {}).get("allocation"):
                        events.append(f"POD {pod_name}: claim {claim_name} not allocatable")
                        continue
                    uid = claim["metadata"]["uid"]
                    msg = dra.Claim(namespace=ns, name=claim_name, uid=uid)
                    r = self.gpu_client.prepare([msg]).claims[uid]
                    if r.error:
                        events.append(f"POD {pod_name}: prepare FAILED: {r.error}")
                        continue
                    dev = claim["status"]["allocation"]["devices"]["results"][0]["device"]
                    self._prepared_pods.setdefault(f"{ns}/{pod_name}", []).append(
                        f"gpu.amd.com:{uid}"
                    )
                    events.append(
                        f"POD {pod_name}: prepared {dev} (shared) -> "
                        f"{r.devices[0].cdi_device_ids[0]}"
                    )
                    continue
                tmpl_name = rc.get("resourceClaimTemplateName")
                if not tmpl_name:
                    continue
                tmpl = None
                deadline0 = time.monotonic() + 10
                while time.monotonic() < deadline0:
                    tmpl = self.client.get_or_none("resourceclaimtemplates", tmpl_name, ns)
                    if tmpl is not None:
                        break
                    time.sleep(0.2)  # the controller may still be rendering it
                if tmpl is None:
                    events.append(f"POD {pod_name}: missing RCT {tmpl_name}")
                    continue
                claim_name = f"{pod_name}-{rc['name']}"
                claim = self.client.apply(
                    "resourceclaims",
                    {
                        "apiVersion": "resource.k8s.io/v1beta1",
                        "kind": "ResourceClaim",
                        "metadata": {"name": claim_name, "namespace": ns},
                        "spec": tmpl["spec"]["spec"],
                    },
                )
                self._pod_claims.setdefault(f"{ns}/{pod_name}", []).append(claim_name)
                deadline = time.monotonic() + self.schedule_timeout
                while time.monotonic() < deadline:
                    self.scheduler.schedule_pending()
                    claim = self.client.get("resourceclaims", claim_name, ns)
                    if (claim.get("status") or {}).get("allocation"):
                        break
                    time.sleep(0.2)
                else:
                    events.append(f"POD {pod_name}: claim {claim_name} unschedulable")
                    continue
                uid = claim["metadata"]["uid"]
                results = claim["status"]["allocation"]["devices"]["results"]
                driver = results[0].get("driver", "gpu.amd.com")
                msg = dra.Claim(namespace=ns, name=claim_name, uid=uid)
                if driver == "gpu.amd.com":
                    resp = self.gpu_client.prepare([msg])
                    r = resp.claims[uid]
                else:
                    r = self.cd_plugin.node_prepare_resources(
                        dra.NodePrepareResourcesRequest(claims=[msg]), None
                    ).claims[uid]
                if r.error:
                    events.append(f"POD {pod_name}: prepare FAILED: {r.error}")
                    continue
                ids = [d.cdi_device_ids[0] for d in r.devices if d.cdi_device_ids]
                cdi_ids.extend(ids)
                self._prepared_pods.setdefault(f"{ns}/{pod_name}", []).append(
                    f"{driver}:{uid}"
                )
                events.append(
                    f"POD {pod_name}: prepared {results[0]['device']} -> {ids[0]}"
                )
            if not pod["spec"].get("resourceClaims"):
                # DRAExtendedResource path: legacy `amd.com/gpu: N` container
                # limits and no claims — the scheduler creates + allocates the
                # special claim and stamps pod.status.extendedResourceClaimStatus
                self.scheduler.schedule_pending()
                live = self.client.get("pods", pod_name, ns)
                ercs = (live.get("status") or {}).get("extendedResourceClaimStatus")
                if ercs:
                    claim_name = ercs["resourceClaimName"]
                    self._pod_claims.setdefault(
                        f"{ns}/{pod_name}", []).append(claim_name)
                    claim = self.client.get("resourceclaims", claim_name, ns)
                    uid = claim["metadata"]["uid"]
                    msg = dra.Claim(namespace=ns, name=claim_name, uid=uid)
                    r = self.gpu_client.prepare([msg]).claims[uid]
                    if r.error:
                        events.append(f"POD {pod_name}: prepare FAILED: {r.error}")
                    else:
                        devs = [res["device"] for res in
                                claim["status"]["allocation"]["devices"]["results"]]
                        self._prepared_pods.setdefault(
                            f"{ns}/{pod_name}", []).append(f"gpu.amd.com:{uid}")
                        events.append(
                            f"POD {pod_name}: prepared {','.join(devs)} "
                            f"(extended-resource) -> {r.devices[0].cdi_device_ids[0]}"
                        )
        return events

    def delete_pod(self, ns: str, pod_name: str) -> None:
        for entry in self._prepared_pods.pop(f"{ns}/{pod_name}", []):
            driver, uid = entry.split(":", 1)
            msg = dra.Claim(uid=uid)
            if driver == "gpu.amd.com":
                self.gpu_client.unprepare([msg])
            else:
                self.cd_plugin.node_unprepare_resources(
                    dra.NodeUnprepareResourcesRequest(claims=[msg]), None
                )
        # GC-lite: per-pod claims (template-generated, extended-resource) are
        # owned by the pod — delete + deallocate so the devices are reusable
        # (shared claims referenced by resourceClaimName are NOT pod-owned and
        # survive). Ref scenario: "ResourceClaim released on pod delete".
        for claim_name in self._pod_claims.pop(f"{ns}/{pod_name}", []):
            claim = self.client.get_or_none("resourceclaims", claim_name, ns)
            if claim is not None:
                self.scheduler.release(claim)
                try:
                    self.client.delete("resourceclaims", claim_name, ns)
                except Exception:
                    pass
        try:
            self.client.delete("pods", pod_name, ns)
        except Exception:
            pass

    def wait_cd_ready(self, name: str, ns: str = "default", timeout: float = 30.0) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            cd = self.client.get_or_none("computedomains", name, ns)
            if cd and (cd.get("status") or {}).get("status") == "Ready":
                return True
            time.sleep(0.2)
        return False
