"""ComputeDomain kubelet plugin: daemon + channel device preparation.

Parity with ``cmd/compute-domain-kubelet-plugin`` (~4.7k LoC Go):

* advertises exactly one ``daemon-0`` and one ``channel-0`` device per node
  (ref nvlib.go:145-193, driver.go:105-120; channels 1..2047 exist internally
  but only 0 is advertised),
* ``ComputeDomainDaemonConfig`` prepare: materialize the per-domain dir
  (``domains/<uid>``) with the fabric daemon config, inject it as
  ``/fabricd`` plus CLIQUE_ID/COMPUTE_DOMAIN_* env via CDI
  (ref device_state.go:594-656),
* ``ComputeDomainChannelConfig`` prepare: assert the CD exists and matches
  the claim namespace, label the node into the CD, **gate on this node being
  Ready in the CD's clique**, then inject the channel (ref :544-592) — the
  MI355X channel is a domain-scoped access token + the shared membership dir
  RCCL bootstrap reads (ROCm has no IMEX channel device),
* co-dependent prepare handled by the retry-with-deadline pattern: the
  kubelet request is retried internally with backoff until ready or the 45 s
  ``ERROR_RETRY_MAX_TIMEOUT`` elapses (ref driver.go:40-60,165-232),
* same checkpoint/cleanup machinery as the GPU plugin.
"""

from __future__ import annotations

import json
import logging
import os
import shutil
import time
from typing import Any, Dict, List, Optional

from .. import API_GROUP, COMPUTE_DOMAIN_DRIVER_NAME
from ..api.configs import (
    ALLOCATION_MODE_ALL,
    ComputeDomainChannelConfig,
    ComputeDomainDaemonConfig,
)
from ..api.decoder import decode_config
from ..api.types import STATUS_READY
from ..cdi.spec import CdiDevice, CdiHandler, ContainerEdits
from ..controller.templates import CD_LABEL_KEY
from ..device.devicelib import DeviceLib
from ..dra import api as dra
from ..k8s.client import Client
from ..plugin.checkpoint import (
    PREPARE_COMPLETED,
    PREPARE_STARTED,
    CheckpointManager,
    ClaimRef,
    PreparedClaim,
    PreparedDevice,
)

logger = logging.getLogger("amddra.cdplugin")

IMEX_CHANNEL_COUNT = 2048  # parity: ref nvlib.go:365-368
ERROR_RETRY_MAX_TIMEOUT = 45.0  # ref driver.go:40-46
RETRY_BASE = 0.1


class PermanentError(RuntimeError):
    """Do not retry (ref driver.go:53-60 permanentError marker)."""


class TransientError(RuntimeError):
    """Retry within the deadline window."""


class ComputeDomainPlugin(dra.DRAPluginServicer):
    def __init__(
        self,
        client: Client,
        devicelib: DeviceLib,
        state_dir: str,
        cdi: Optional[CdiHandler] = None,
        node_name: str = "",
        driver_name: str = COMPUTE_DOMAIN_DRIVER_NAME,
        retry_max_timeout: float = ERROR_RETRY_MAX_TIMEOUT,
        strict_fabric: bool = False,
    ):
        self.client = client
        self.devicelib = devicelib
        self.node_name = node_name or os.environ.get("NODE_NAME", "node")
        self.driver_name = driver_name
        self.state_dir = state_dir
        os.makedirs(state_dir, exist_ok=True)
        self.domains_dir = os.path.join(state_dir, "domains")
        os.makedirs(self.domains_dir, exist_ok=True)
        self.cdi = cdi or CdiHandler(
            cdi_root=os.path.join(state_dir, "cdi"),
            vendor="amd.com",
            klass="compute-domain",
            driver_name=driver_name,
        )
        self.checkpoints = CheckpointManager(state_dir)
        self.retry_max_timeout = retry_max_timeout
        # CrashOnXGMIFabricErrors gate (ref nvlib.go strict mode vs legacy
        # fallback): with strict_fabric, a GPU node without a derivable xGMI
        # clique is a hard error instead of clique-less operation.
        self.strict_fabric = strict_fabric
        # Boot-time "standard" management spec: full driver injection under
        # device name `all`; per-claim specs carry only config-state edits
        # (ref cdi.go:142-206). Written once per plugin start.
        try:
            self.cdi.write_standard_spec(
                sorted(g.render_minor for g in self.devicelib.gpus())
            )
        except Exception:
            logger.exception("failed to write boot-time standard CDI spec")

    # ------------------------------------------------------------------
    # Device model: daemon-0 + channel-0
    # ------------------------------------------------------------------

    def resource_slice(self) -> Dict[str, Any]:
        devices = [
            {
                "name": "daemon-0",
                "basic": {
                    "attributes": {"type": {"string": "daemon"}, "id": {"int": 0}},
                    "capacity": {},
                },
            },
            {
                "name": "channel-0",
                "basic": {
                    "attributes": {"type": {"string": "channel"}, "id": {"int": 0}},
                    "capacity": {},
                },
            },
        ]
        return {
            "apiVersion": "resource.k8s.io/v1beta1",
            "kind": "ResourceSlice",
            "metadata": {
                "name": f"{self.node_name}-{self.driver_name.replace('.', '-')}",
                "labels": {f"{API_GROUP}/node": self.node_name},
            },
            "spec": {
                "driver": self.driver_name,
                "nodeName": self.node_name,
                "pool": {"name": self.node_name, "resourceSliceCount": 1, "generation": 1},
                "devices": devices,
            },
        }

    # ------------------------------------------------------------------
    # Clique identity (ref nvlib.go:195-363 getCliqueID)
    # ------------------------------------------------------------------

    def clique_id(self) -> str:
        topo = self.devicelib.topology()
        gpus = self.devicelib.gpus()
        if not gpus:
            if self.strict_fabric:
                raise RuntimeError("strict fabric mode: no GPUs enumerated")
            return ""
        cid = topo.clique_id_for(gpus[0].uuid)
        if not cid and self.strict_fabric and len(gpus) > 1:
            raise RuntimeError(
                "strict fabric mode: multi-GPU node without an xGMI hive"
            )
        return cid

    # ------------------------------------------------------------------
    # gRPC entry points with retry-with-deadline
    # ------------------------------------------------------------------

    def node_prepare_resources(self, req, context):
        resp = dra.NodePrepareResourcesResponse()
        for claim in req.claims:
            resp.claims[claim.uid] = self._with_retry(
                lambda c=claim: self._prepare_claim(c),
                lambda err: dra.NodePrepareResourceResponse(error=err),
            )
        return resp

    def node_unprepare_resources(self, req, context):
        resp = dra.NodeUnprepareResourcesResponse()
        for claim in req.claims:
            try:
                self._unprepare_claim(claim.uid)
                resp.claims[claim.uid] = dra.NodeUnprepareResourceResponse()
            except Exception as e:
                logger.exception("cd unprepare failed for %s", claim.uid)
                resp.claims[claim.uid] = dra.NodeUnprepareResourceResponse(error=str(e))
        return resp

    def _with_retry(self, fn, err_result):
        deadline = time.monotonic() + self.retry_max_timeout
        delay = RETRY_BASE
        while True:
            try:
                return fn()
            except PermanentError as e:
                return err_result(str(e))
            except Exception as e:
                if time.monotonic() + delay >= deadline:
                    return err_result(f"retry window exhausted: {e}")
                time.sleep(delay)
                delay = min(delay * 2, 6.0)

    # ------------------------------------------------------------------
    # Prepare
    # ------------------------------------------------------------------

    def _resolve(self, namespace: str, name: str, uid: str):
        """Fetch the ResourceClaim's allocation + opaque config."""
        claim = self.client.get_or_none("resourceclaims", name, namespace)
        if claim is None:
            raise TransientError(f"resourceclaim {namespace}/{name} not found")
        alloc = ((claim.get("status") or {}).get("allocation") or {})
        results = ((alloc.get("devices") or {}).get("results")) or []
        configs = ((alloc.get("devices") or {}).get("config")) or []
        return claim, results, configs

    def _prepare_claim(self, claim_msg) -> dra.NodePrepareResourceResponse:
        """Two-phase prepare (ref device_state.go:186-256):

        1. durable ``PrepareStarted`` intent BEFORE any side effect, with
           channel exclusivity asserted atomically inside the RMW;
        2. side effects (domain dir, node label, readiness gate, CDI spec);
        3. ``PrepareCompleted`` commit, exclusivity re-asserted atomically.

        A crash between 1 and 3 leaves a ``PrepareStarted`` entry that
        annotates the potentially-partial prepare: a retry re-enters and
        finishes it, and the async cleanup manager reaps it if the claim
        goes stale (ref device_state.go:239-243, cleanup.go:117-125).
        A PermanentError after phase 1 (e.g. losing a channel race) rolls
        back this claim's side effects."""
        uid = claim_msg.uid
        existing = self.checkpoints.load().get_claim(uid)
        if existing is not None and existing.state == PREPARE_COMPLETED:
            return dra.NodePrepareResourceResponse(
                devices=[
                    dra.Device(
                        pool_name=self.node_name,
                        device_name=d.name,
                        cdi_device_ids=d.cdi_device_ids,
                    )
                    for d in existing.devices
                ]
            )
        _, results, configs = self._resolve(claim_msg.namespace, claim_msg.name, uid)
        intents = []  # (device_name, cfg)
        for res in results:
            if res.get("driver") not in (None, self.driver_name):
                continue
            device_name = res.get("device", "")
            cfg = self._config_for(res, configs)
            if not isinstance(cfg, (ComputeDomainDaemonConfig, ComputeDomainChannelConfig)):
                raise PermanentError(
                    f"device {device_name} has no ComputeDomain opaque config"
                )
            intents.append((device_name, cfg))
        ref = ClaimRef(namespace=claim_msg.namespace, name=claim_msg.name, uid=uid)

        # Phase 1: PrepareStarted — durable intent with the would-be devices,
        # channel reservation checked atomically in the same RMW.
        def start(data):
            intent_devices = []
            for device_name, cfg in intents:
                if isinstance(cfg, ComputeDomainChannelConfig):
                    self._check_channel_free(data, uid, cfg.domain_id, 0)
                    intent_devices.append(
                        PreparedDevice(
                            type="channel", name=device_name, uuid=cfg.domain_id
                        )
                    )
                else:
                    intent_devices.append(
                        PreparedDevice(
                            type="daemon", name=device_name, uuid=cfg.domain_id
                        )
                    )
            data.set_claim(
                uid,
                PreparedClaim(state=PREPARE_STARTED, claim=ref, devices=intent_devices),
            )

        self.checkpoints.update(start)

        created_dirs: List[str] = []
        try:
            devices: List[PreparedDevice] = []
            cdi_devices: List[CdiDevice] = []
            out_devices: List[dra.Device] = []
            for device_name, cfg in intents:
                if isinstance(cfg, ComputeDomainDaemonConfig):
                    pd, cd_dev = self._prepare_daemon(
                        claim_msg, device_name, cfg, created_dirs
                    )
                else:
                    pd, cd_dev = self._prepare_channel(
                        claim_msg, device_name, cfg, created_dirs
                    )
                devices.append(pd)
                cdi_devices.append(cd_dev)
            cdi_ids = self.cdi.write_claim_spec(uid, cdi_devices)
            for pd, cid in zip(devices, cdi_ids):
                ids = [cid]
                if pd.type == "daemon":
                    # daemon containers also get the boot-time standard
                    # (management) device for full driver injection
                    # (ref device_state.go:467 GetStandardDevice — empty for
                    # channel-type devices)
                    ids = [self.cdi.qualified_name("all"), cid]
                pd.cdi_device_ids = ids
                out_devices.append(
                    dra.Device(
                        pool_name=self.node_name, device_name=pd.name, cdi_device_ids=ids
                    )
                )

            # Phase 2: PrepareCompleted — exclusivity re-asserted atomically
            # inside the commit RMW (multi-process safe, strictly stronger
            # than the reference's in-process lock;
            # ref assertImexChannelNotAllocated device_state.go:729-757).
            def commit(data):
                for pd in devices:
                    if pd.type == "channel":
                        self._check_channel_free(
                            data, uid, pd.uuid, pd.partition_index
                        )
                data.set_claim(
                    uid,
                    PreparedClaim(state=PREPARE_COMPLETED, claim=ref, devices=devices),
                )

            self.checkpoints.update(commit)
            return dra.NodePrepareResourceResponse(devices=out_devices)
        except PermanentError:
            self._rollback_started(uid, created_dirs)
            raise
        # TransientErrors propagate with the PrepareStarted entry left in
        # place (deliberate): the retry loop re-enters, and stale entries are
        # reaped by cleanup_stale_claims.

    def _check_channel_free(
        self, data, claim_uid: str, domain_id: str, channel: int = 0
    ) -> None:
        """Channel exclusivity against COMPLETED claims only: a claim stuck
        in PrepareStarted is either about to retry (we win the race and it
        re-checks) or stale (cleanup reaps it) — ref device_state.go:736-745."""
        for ouid, pc in data.claims().items():
            if ouid == claim_uid or pc is None or pc.state != PREPARE_COMPLETED:
                continue
            for od in pc.devices or []:
                if (
                    od.type == "channel"
                    and od.uuid == domain_id
                    and od.partition_index == channel
                ):
                    raise PermanentError(
                        f"channel {channel} of domain {domain_id} "
                        f"already allocated to claim {ouid}"
                    )

    def _rollback_started(self, uid: str, created_dirs: List[str]) -> None:
        """Undo a failed prepare's side effects: claim CDI spec, checkpoint
        entry, and any domain dir THIS prepare created that no surviving
        claim references."""
        try:
            self.cdi.delete_claim_spec(uid)
        except Exception:
            logger.exception("rollback: CDI spec removal for %s failed", uid)
        try:
            self.checkpoints.update(lambda d: d.remove_claim(uid))
        except Exception:
            logger.exception("rollback: checkpoint removal for %s failed", uid)
        if not created_dirs:
            return
        cp = self.checkpoints.load()
        referenced = set()
        for pc in cp.claims().values():
            for d in (pc.devices if pc else []) or []:
                referenced.add(d.uuid)
        for ddir in created_dirs:
            if os.path.basename(ddir) not in referenced:
                shutil.rmtree(ddir, ignore_errors=True)

    def _config_for(self, result: Dict[str, Any], configs: List[Dict[str, Any]]):
        request = result.get("request", "")
        chosen = None
        for c in configs:
            opaque = (c.get("opaque") or {})
            if opaque.get("driver") != self.driver_name:
                continue
            reqs = c.get("requests") or []
            if not reqs or request in reqs:
                chosen = opaque.get("parameters")
        if chosen is None:
            raise PermanentError(f"no opaque config for request {request!r}")
        cfg = decode_config(chosen, strict=True)
        cfg.normalize()
        cfg.validate()
        return cfg

    # -- daemon device ------------------------------------------------------

    def domain_dir(self, cd_uid: str) -> str:
        return os.path.join(self.domains_dir, cd_uid)

    def _ensure_domain_dir(self, cd_uid: str, created_dirs: List[str]) -> str:
        """Create the per-domain dir, recording it for rollback only when
        THIS call created it (a pre-existing dir belongs to other claims)."""
        ddir = self.domain_dir(cd_uid)
        if not os.path.isdir(ddir):
            created_dirs.append(ddir)
        os.makedirs(os.path.join(ddir, "shared"), exist_ok=True)
        return ddir

    def _prepare_daemon(
        self,
        claim_msg,
        device_name: str,
        cfg: ComputeDomainDaemonConfig,
        created_dirs: List[str],
    ):
        cd = self._get_cd_by_uid(cfg.domain_id)
        if cd is None:
            raise TransientError(f"ComputeDomain {cfg.domain_id} not found")
        ddir = self._ensure_domain_dir(cfg.domain_id, created_dirs)
        clique = self.clique_id()
        # config consumed by the fabric daemon supervisor (imexd.cfg analog)
        with open(os.path.join(ddir, "fabricd.cfg.template"), "w") as f:
            json.dump(
                {
                    "domain": cfg.domain_id,
                    "cliqueID": clique,
                    "peerPort": 50000,
                    "commandPort": 50005,
                    "nodesConfig": "nodes.cfg",
                },
                f,
                indent=2,
            )
        edits = ContainerEdits(
            env=[
                f"CLIQUE_ID={clique}",
                f"COMPUTE_DOMAIN_UUID={cfg.domain_id}",
                f"COMPUTE_DOMAIN_NAME={cd['metadata']['name']}",
                f"COMPUTE_DOMAIN_NAMESPACE={cd['metadata']['namespace']}",
                f"NODE_NAME={self.node_name}",
            ],
            mounts=[
                {
                    "hostPath": ddir,
                    "containerPath": "/fabricd",
                    "options": ["rw", "bind"],
                }
            ],
        )
        # fabric devices are injected only when a clique exists
        # (ref device_state.go:648-652 gates the imex-mgmt node on cliqueID)
        if clique:
            edits.device_nodes.append(self.cdi.kfd_node())
        cd_dev = CdiDevice(name=f"claim-{claim_msg.uid}-{device_name}", edits=edits)
        pd = PreparedDevice(type="daemon", name=device_name, uuid=cfg.domain_id,
                            device_nodes=[n.path for n in edits.device_nodes])
        return pd, cd_dev

    # -- channel device ------------------------------------------------------

    def _prepare_channel(
        self,
        claim_msg,
        device_name: str,
        cfg: ComputeDomainChannelConfig,
        created_dirs: List[str],
    ):
        cd = self._get_cd_by_uid(cfg.domain_id)
        if cd is None:
            raise TransientError(f"ComputeDomain {cfg.domain_id} not found")
        if cd["metadata"]["namespace"] != claim_msg.namespace:
            raise PermanentError(
                f"claim namespace {claim_msg.namespace} does not match ComputeDomain "
                f"namespace {cd['metadata']['namespace']}"
            )
        self._label_node(cfg.domain_id)
        self._assert_domain_ready_on_node(cd)
        ddir = self._ensure_domain_dir(cfg.domain_id, created_dirs)
        if cfg.allocation_mode == ALLOCATION_MODE_ALL:
            channels = list(range(IMEX_CHANNEL_COUNT))
        else:
            channels = [0]
        edits = ContainerEdits(
            env=[
                f"COMPUTE_DOMAIN_UUID={cfg.domain_id}",
                f"COMPUTE_DOMAIN_CHANNELS={','.join(map(str, channels[:8]))}"
                + ("..." if len(channels) > 8 else ""),
                f"COMPUTE_DOMAIN_CHANNEL_COUNT={len(channels)}",
            ],
            mounts=[
                {
                    "hostPath": os.path.join(ddir, "shared"),
                    "containerPath": "/compute-domain",
                    "options": ["rw", "bind"],
                }
            ],
        )
        cd_dev = CdiDevice(name=f"claim-{claim_msg.uid}-{device_name}", edits=edits)
        pd = PreparedDevice(
            type="channel",
            name=device_name,
            uuid=cfg.domain_id,
            partition_index=0,
            device_nodes=[],
        )
        return pd, cd_dev

    def _label_node(self, cd_uid: str) -> None:
        node = self.client.get_or_none("nodes", self.node_name)
        if node is None:
            self.client.create(
                "nodes",
                {"apiVersion": "v1", "kind": "Node",
                 "metadata": {"name": self.node_name, "labels": {CD_LABEL_KEY: cd_uid}}},
            )
            return
        labels = node["metadata"].get("labels") or {}
        if labels.get(CD_LABEL_KEY) != cd_uid:
            self.client.patch(
                "nodes", self.node_name, {"metadata": {"labels": {CD_LABEL_KEY: cd_uid}}}
            )

    def _assert_domain_ready_on_node(self, cd: Dict[str, Any]) -> None:
        """Channel prepare gates on THIS node being Ready in the clique
        (ref computedomain.go:198-236 AssertComputeDomainReady)."""
        uid = cd["metadata"]["uid"]
        for clique in self.client.list("computedomaincliques"):
            if not clique["metadata"]["name"].startswith(uid + "."):
                continue
            for d in clique.get("daemons") or []:
                if d.get("nodeName") == self.node_name:
                    if d.get("status") == STATUS_READY:
                        return
                    raise TransientError(
                        f"daemon on node {self.node_name} not Ready in clique"
                    )
        # no clique entry for this node: if the node has no clique (no xGMI
        # hive), fall back to CD-level status (ref cdstatus non-fabric path)
        if not self.clique_id():
            if ((cd.get("status") or {}).get("status")) == STATUS_READY:
                return
        raise TransientError(f"node {self.node_name} not registered in domain {uid}")

    # ------------------------------------------------------------------
    # Unprepare / cleanup
    # ------------------------------------------------------------------

    def _unprepare_claim(self, uid: str) -> None:
        self.cdi.delete_claim_spec(uid)
        self.checkpoints.update(lambda d: d.remove_claim(uid))

    def cleanup_stale_domain_dirs(self) -> int:
        """Remove per-CD dirs whose CD no longer exists (ref
        computedomain.go:384-439, 10-min cadence)."""
        live = {cd["metadata"]["uid"] for cd in self.client.list("computedomains")}
        removed = 0
        for entry in os.listdir(self.domains_dir):
            if entry not in live:
                shutil.rmtree(os.path.join(self.domains_dir, entry), ignore_errors=True)
                removed += 1
        return removed

    def cleanup_stale_claims(self) -> int:
        """Unprepare checkpointed claims whose ResourceClaim is gone or has a
        different UID (the GPU plugin's CheckpointCleanupManager analog, ref
        compute-domain-kubelet-plugin/cleanup.go), plus orphaned CDI specs."""
        removed = 0
        cp = self.checkpoints.load()
        for uid, pc in cp.claims().items():
            if pc is None:
                continue
            ref = pc.claim
            obj = self.client.get_or_none("resourceclaims", ref.name, ref.namespace)
            if obj is not None and obj.get("metadata", {}).get("uid") == uid:
                continue
            logger.info("cd cleanup: claim %s gone; unpreparing", uid)
            try:
                self._unprepare_claim(uid)
                removed += 1
            except Exception:
                logger.exception("cd cleanup: unprepare of %s failed", uid)
        live = set(self.checkpoints.load().claims())
        for uid in self.cdi.list_claim_uids():
            if uid not in live:
                self.cdi.delete_claim_spec(uid)
                removed += 1
        return removed

    def _get_cd_by_uid(self, uid: str) -> Optional[Dict[str, Any]]:
        for cd in self.client.list("computedomains"):
            if cd["metadata"]["uid"] == uid:
                return cd
        return None
