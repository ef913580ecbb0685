"""CDI (Container Device Interface) spec generation for ROCm devices.

Parity with the reference's CDI layer (``cmd/gpu-kubelet-plugin/cdi.go``:
per-claim transient spec files written to ``/var/run/cdi`` with a 5-min
expiring cache at :112-169, ``CreateClaimSpecFile`` at :181-306) — built for
the ROCm runtime instead of ``nvidia-container-toolkit``:

* devices are ``/dev/kfd`` (compute) + ``/dev/dri/renderD*`` (+ card node),
* driver enablement is a set of library mounts from the ROCm root plus an
  ``update-ldcache`` hook equivalent (we mount a generated ld.so.conf drop-in
  instead of executing a hook binary — hookless, works on any runtime),
* env: ``AMD_VISIBLE_DEVICES``-style enumeration plus ROCm selector env
  (``ROCR_VISIBLE_DEVICES`` for KFD-level isolation by GPU index/UUID).

Spec files are JSON per CDI 0.6.0.  The writer is transactional (tmp+rename)
and idempotent; ``delete_claim_spec`` is a no-op when absent.
"""

from __future__ import annotations

import json
import os
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from .. import GPU_DRIVER_NAME

CDI_VERSION = "0.6.0"
DEFAULT_CDI_ROOT = "/var/run/cdi"
SPEC_CACHE_TTL = 300.0  # 5 min (ref cdi.go:132,165)


@dataclass
class DeviceNode:
    path: str
    host_path: str = ""
    type: str = "c"
    permissions: str = "rw"

    def render(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {"path": self.path}
        if self.host_path and self.host_path != self.path:
            out["hostPath"] = self.host_path
        if self.type:
            out["type"] = self.type
        if self.permissions:
            out["permissions"] = self.permissions
        return out


@dataclass
class ContainerEdits:
    device_nodes: List[DeviceNode] = field(default_factory=list)
    env: List[str] = field(default_factory=list)
    mounts: List[Dict[str, Any]] = field(default_factory=list)
    hooks: List[Dict[str, Any]] = field(default_factory=list)

    def render(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {}
        if self.device_nodes:
            out["deviceNodes"] = [d.render() for d in self.device_nodes]
        if self.env:
            out["env"] = list(self.env)
        if self.mounts:
            out["mounts"] = list(self.mounts)
        if self.hooks:
            out["hooks"] = list(self.hooks)
        return out


@dataclass
class CdiDevice:
    name: str
    edits: ContainerEdits = field(default_factory=ContainerEdits)

    def render(self) -> Dict[str, Any]:
        return {"name": self.name, "containerEdits": self.edits.render()}


class CdiHandler:
    """Writes per-claim transient CDI specs + a node-common spec."""

    def __init__(
        self,
        cdi_root: str = "",
        vendor: str = "amd.com",
        klass: str = "gpu",
        dev_root: str = "",
        rocm_root: str = "/opt/rocm",
        driver_name: str = GPU_DRIVER_NAME,
    ):
        self.cdi_root = cdi_root or os.environ.get("AMDDRA_CDI_ROOT", DEFAULT_CDI_ROOT)
        self.vendor = vendor
        self.klass = klass
        self.dev_root = dev_root or os.environ.get("AMDDRA_DEV_ROOT", "/dev")
        self.rocm_root = rocm_root
        self.driver_name = driver_name
        os.makedirs(self.cdi_root, exist_ok=True)
        self._cache: Dict[str, float] = {}
        self._cache_lock = threading.Lock()

    # -- naming ------------------------------------------------------------

    def qualified_name(self, device: str) -> str:
        return f"{self.vendor}/{self.klass}={device}"

    def claim_spec_path(self, claim_uid: str) -> str:
        return os.path.join(self.cdi_root, f"{self.driver_name}-claim-{claim_uid}.json")

    def common_spec_path(self) -> str:
        return os.path.join(self.cdi_root, f"{self.driver_name}-common.json")

    def standard_spec_path(self) -> str:
        return os.path.join(self.cdi_root, f"{self.driver_name}-standard.json")

    # -- device-node edits ---------------------------------------------------

    def _host(self, path: str) -> str:
        """Map a canonical /dev path to the (possibly re-rooted) host path."""
        if path.startswith("/dev/") and self.dev_root != "/dev":
            return os.path.join(self.dev_root, path[len("/dev/"):])
        return path

    def kfd_node(self) -> DeviceNode:
        return DeviceNode(path="/dev/kfd", host_path=self._host("/dev/kfd"))

    def render_node(self, render_minor: int) -> DeviceNode:
        p = f"/dev/dri/renderD{render_minor}"
        return DeviceNode(path=p, host_path=self._host(p))

    def card_node(self, card_minor: int) -> DeviceNode:
        p = f"/dev/dri/card{card_minor}"
        return DeviceNode(path=p, host_path=self._host(p))

    def gpu_edits(
        self,
        render_minors: List[int],
        card_minors: Optional[List[int]] = None,
        env: Optional[List[str]] = None,
    ) -> ContainerEdits:
        edits = ContainerEdits()
        edits.device_nodes.append(self.kfd_node())
        for rm in render_minors:
            edits.device_nodes.append(self.render_node(rm))
        for cm in card_minors or []:
            edits.device_nodes.append(self.card_node(cm))
        if env:
            edits.env.extend(env)
        return edits

    # -- spec writing --------------------------------------------------------

    def write_claim_spec(self, claim_uid: str, devices: List[CdiDevice]) -> List[str]:
        """Write the per-claim transient spec; returns fully-qualified CDI ids
        (ref CreateClaimSpecFile, cdi.go:181-306)."""
        spec = {
            "cdiVersion": CDI_VERSION,
            "kind": f"{self.vendor}/{self.klass}",
            "annotations": {
                f"{self.vendor}/claim-uid": claim_uid,
                f"{self.vendor}/driver": self.driver_name,
            },
            "devices": [d.render() for d in devices],
        }
        path = self.claim_spec_path(claim_uid)
        with self._cache_lock:
            cached = self._cache.get(path)
            if cached is not None and time.monotonic() - cached < SPEC_CACHE_TTL:
                if os.path.exists(path):
                    existing = json.load(open(path))
                    if existing == spec:
                        return [self.qualified_name(d.name) for d in devices]
        tmp = path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            json.dump(spec, f, indent=2, sort_keys=True)
        os.replace(tmp, path)
        with self._cache_lock:
            self._cache[path] = time.monotonic()
        return [self.qualified_name(d.name) for d in devices]

    def delete_claim_spec(self, claim_uid: str) -> None:
        path = self.claim_spec_path(claim_uid)
        try:
            os.unlink(path)
        except FileNotFoundError:
            pass
        with self._cache_lock:
            self._cache.pop(path, None)

    def write_common_spec(self) -> str:
        """Node-common boot-time spec with the ROCm driver-enablement edits
        (the nvidia-cdi-hook/driver-mount analog; ref
        compute-domain-kubelet-plugin/cdi.go:142-206 'standard' spec)."""
        libs = self._rocm_runtime_mounts()
        spec = {
            "cdiVersion": CDI_VERSION,
            "kind": f"{self.vendor}/{self.klass}",
            "devices": [
                {
                    "name": "common",
                    "containerEdits": {
                        "deviceNodes": [self.kfd_node().render()],
                        "env": [f"AMDDRA_DRIVER={self.driver_name}"],
                        "mounts": libs,
                    },
                }
            ],
        }
        path = self.common_spec_path()
        tmp = path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            json.dump(spec, f, indent=2, sort_keys=True)
        os.replace(tmp, path)
        return path

    def write_standard_spec(self, render_minors: Optional[List[int]] = None) -> str:
        """Boot-time "standard" management spec: one device named ``all``
        carrying the full driver injection (/dev/kfd, every render node, ROCm
        runtime mounts). Daemon-type prepared devices reference
        ``<vendor>/<klass>=all`` IN ADDITION to their per-claim device, so the
        fabric-daemon container gets driver enablement even though the
        per-claim spec carries only config-state edits (ref
        compute-domain-kubelet-plugin/cdi.go:142-206 CreateStandardDeviceSpecFile,
        device_state.go:467 GetStandardDevice)."""
        edits = ContainerEdits()
        edits.device_nodes.append(self.kfd_node())
        for rm in render_minors or []:
            edits.device_nodes.append(self.render_node(rm))
        edits.env.append(f"AMDDRA_DRIVER={self.driver_name}")
        # keep lower-level runtimes from double-injecting devices
        # (the reference sets NVIDIA_VISIBLE_DEVICES=void for the same reason)
        edits.env.append("AMD_VISIBLE_DEVICES=void")
        spec = {
            "cdiVersion": CDI_VERSION,
            "kind": f"{self.vendor}/{self.klass}",
            "devices": [CdiDevice(name="all", edits=edits).render()],
        }
        spec["devices"][0]["containerEdits"]["mounts"] = self._rocm_runtime_mounts()
        path = self.standard_spec_path()
        tmp = path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            json.dump(spec, f, indent=2, sort_keys=True)
        os.replace(tmp, path)
        return path

    def _rocm_runtime_mounts(self) -> List[Dict[str, Any]]:
        """Mount the minimal ROCm user-space runtime (HIP, HSA, RCCL) into the
        container when present on the host."""
        mounts = []
        candidates = [
            "lib/libamdhip64.so",
            "lib/libhsa-runtime64.so",
            "lib/librccl.so",
            "lib/librocm_smi64.so",
            "lib/libamd_smi.so",
        ]
        for rel in candidates:
            host = os.path.join(self.rocm_root, rel)
            if os.path.exists(host):
                mounts.append(
                    {
                        "hostPath": host,
                        "containerPath": host,
                        "options": ["ro", "nosuid", "nodev", "bind"],
                    }
                )
        return mounts

    # -- maintenance ---------------------------------------------------------

    def list_claim_uids(self) -> List[str]:
        prefix = f"{self.driver_name}-claim-"
        uids = []
        for name in os.listdir(self.cdi_root):
            if name.startswith(prefix) and name.endswith(".json"):
                uids.append(name[len(prefix):-len(".json")])
        return uids
