"""Versioned on-disk checkpoint with two-phase prepare states.

Parity with the reference's checkpoint subsystem
(``cmd/gpu-kubelet-plugin/checkpoint.go:26-145``, ``checkpointv.go:29-136``,
``device_state.go:590-676``):

* one JSON file (``checkpoint.json``) holding **both** V1 and V2 payloads with
  per-version checksums so up- and down-grades can each validate the payload
  they understand,
* claims move through ``PrepareStarted`` -> ``PrepareCompleted`` (durable
  intent, enabling partial-prepare rollback after a crash),
* ``nodeBootID`` invalidation: device state (partition modes) does not
  survive a reboot, so prepared claims recorded under a different boot id are
  discarded (ref ``device_state.go:186-226``),
* all mutation is a read-modify-write under an exclusive flock
  (``cp.lock``), shared with any other driver-pod instance on the node,
* checksum mismatch raises with a diagnostic including a unified diff of the
  stored vs recomputed canonical form (ref ``device_state.go:611-640``).

The checksum is CRC-stable by construction: canonical JSON (sorted keys,
compact separators) of the payload with the checksum field zeroed, and
``omitempty`` serialization discipline everywhere (``serde.to_dict``).
"""

from __future__ import annotations

import difflib
import json
import os
import zlib
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..api.serde import api_field, from_dict, to_dict
from ..utils.bootid import read_boot_id
from ..utils.flock import Flock

PREPARE_STARTED = "PrepareStarted"
PREPARE_COMPLETED = "PrepareCompleted"

CHECKPOINT_FILE = "checkpoint.json"
CHECKPOINT_LOCK = "cp.lock"


class CheckpointCorrupt(RuntimeError):
    pass


@dataclass
class ClaimRef:
    namespace: str = api_field("namespace", default="")
    name: str = api_field("name", default="")
    uid: str = api_field("uid", default="")

    def __str__(self) -> str:  # canonical claim string (ref types.go:48-70)
        return f"{self.namespace}/{self.name}:{self.uid}"


@dataclass
class PreparedDevice:
    """Union entry (ref prepared.go:31-122): exactly one of gpu / partition /
    vfio / channel / daemon is populated (by `type`)."""

    type: str = api_field("type", default="gpu")  # gpu|partition|vfio|channel|daemon
    name: str = api_field("name", default="")  # canonical device name (gpu-0, gpu-0-cpx-3)
    request: str = api_field("request", default="")  # claim request satisfied
    uuid: str = api_field("uuid", default="")
    parent_uuid: str = api_field("parentUUID", default="")
    compute_mode: str = api_field("computeMode", default="")
    memory_mode: str = api_field("memoryMode", default="")
    partition_index: int = api_field("partitionIndex", default=0)
    # PCI bus ID, checkpointed for type=vfio: a GPU bound to vfio-pci has no
    # drm card so UUID lookup fails after a plugin restart — unbind must use
    # the address directly (ref prepared.go checkpoints PciBusID for Vfio).
    # Optional so existing gpu/partition payloads keep identical bytes.
    pci_bus_id: Optional[str] = api_field("pciBusID", default=None)
    cdi_device_ids: List[str] = api_field("cdiDeviceIDs", default_factory=list)
    device_nodes: List[str] = api_field("deviceNodes", default_factory=list)
    config: Optional[Dict[str, Any]] = api_field("config", default=None)  # opaque config used


@dataclass
class PreparedClaim:
    state: str = api_field("state", default=PREPARE_STARTED)
    claim: ClaimRef = api_field("claim", default_factory=ClaimRef)
    devices: List[PreparedDevice] = api_field("devices", default_factory=list)


@dataclass
class CheckpointData:
    node_boot_id: str = api_field("nodeBootID", default="")
    prepared_claims: Dict[str, Any] = api_field("preparedClaims", default_factory=dict)

    def get_claim(self, uid: str) -> Optional[PreparedClaim]:
        raw = self.prepared_claims.get(uid)
        if raw is None:
            return None
        if isinstance(raw, PreparedClaim):
            return raw
        return from_dict(PreparedClaim, raw, strict=False)

    def set_claim(self, uid: str, claim: PreparedClaim) -> None:
        raw = to_dict(claim)
        self.prepared_claims[uid] = raw
        self._frags()[uid] = f"{json.dumps(uid)}:{_canonical(raw)}"

    def remove_claim(self, uid: str) -> None:
        self.prepared_claims.pop(uid, None)
        self._frags().pop(uid, None)

    def claims(self) -> Dict[str, PreparedClaim]:
        return {uid: self.get_claim(uid) for uid in self.prepared_claims}

    # -- canonical-payload composition ----------------------------------
    # Claim entries are immutable once set, so their canonical JSON
    # fragments are cached: a store costs O(changed claims) to serialize
    # instead of O(all claims). Byte-identical to
    # _canonical(to_dict(self)) including omitempty semantics (verified by
    # the checksum-stability tests).

    def _frags(self) -> Dict[str, str]:
        f = getattr(self, "_frag_cache", None)
        if f is None:
            f = {}
            object.__setattr__(self, "_frag_cache", f)
        return f

    def canonical_payload(self) -> str:
        # Fragments are cached as complete '"<uid>":<claim-json>' strings so
        # a store with N standing claims is a sorted join of cached strings
        # (one fresh serialization for the mutated claim only) — the
        # whole-checkpoint-rewrite cost at large populations is IO+checksum,
        # not re-serialization (measured: 2000 standing claims).
        frags = self._frags()
        parts = []
        if self.node_boot_id:
            parts.append(f'"nodeBootID":{json.dumps(self.node_boot_id)}')
        if self.prepared_claims:
            entries = []
            for uid in sorted(self.prepared_claims):
                frag = frags.get(uid)
                if frag is None:
                    body = _canonical(to_dict(self.prepared_claims[uid])
                                      if not isinstance(self.prepared_claims[uid], dict)
                                      else self.prepared_claims[uid])
                    frag = f"{json.dumps(uid)}:{body}"
                    frags[uid] = frag
                entries.append(frag)
            parts.append('"preparedClaims":{' + ",".join(entries) + "}")
        return "{" + ",".join(parts) + "}"


def _canonical(obj: Any) -> str:
    return json.dumps(obj, sort_keys=True, separators=(",", ":"))


def _checksum(payload: Dict[str, Any]) -> int:
    return zlib.crc32(_canonical(payload).encode("utf-8")) & 0xFFFFFFFF


class CheckpointManager:
    """Flock-guarded read-modify-write over the dual-version checkpoint file."""

    SUPPORTED_VERSIONS = ("v1", "v2")
    WRITE_VERSIONS = ("v1", "v2")  # dual-write for downgrade safety

    def __init__(self, state_dir: str, boot_id: str = ""):
        self.state_dir = state_dir
        os.makedirs(state_dir, exist_ok=True)
        self.path = os.path.join(state_dir, CHECKPOINT_FILE)
        self.lock = Flock(os.path.join(state_dir, CHECKPOINT_LOCK))
        self.boot_id = boot_id if boot_id != "" else read_boot_id()
        # Read cache, valid only while the file's (ino, size, mtime_ns) is
        # unchanged. Safe because every mutation (ours or another driver
        # pod's) happens under the flock and replaces the file atomically.
        self._cache_stat = None
        self._cache_payload: str = ""
        # parsed-object cache used ONLY by update(): avoids re-parsing the
        # whole checkpoint on every RMW of a hot prepare path; invalidated
        # whenever a mutate raises mid-flight or the file changes on disk
        self._cache_obj: "CheckpointData | None" = None

    # -- raw IO ------------------------------------------------------------

    def _stat_key(self):
        try:
            st = os.stat(self.path)
            return (st.st_ino, st.st_size, st.st_mtime_ns)
        except OSError:
            return None

    def _load_unlocked(self) -> CheckpointData:
        if not os.path.exists(self.path):
            return CheckpointData(node_boot_id=self.boot_id)
        key = self._stat_key()
        if key is not None and key == self._cache_stat and self._cache_payload:
            data = from_dict(
                CheckpointData, json.loads(self._cache_payload), strict=False
            )
            data.node_boot_id = data.node_boot_id or self.boot_id
            return data
        with open(self.path, "r", encoding="utf-8") as f:
            try:
                raw = json.load(f)
            except json.JSONDecodeError as e:
                raise CheckpointCorrupt(f"checkpoint is not valid JSON: {e}") from None
        data = self._extract_version(raw)
        if data.node_boot_id and data.node_boot_id != self.boot_id:
            # Reboot: partition/device state did not survive; prepared claims
            # are stale (ref device_state.go:186-226).
            return CheckpointData(node_boot_id=self.boot_id)
        data.node_boot_id = data.node_boot_id or self.boot_id
        return data

    def _extract_version(self, raw: Dict[str, Any]) -> CheckpointData:
        # Legacy (pre-versioning) flat format: {"nodeBootID", "preparedClaims"}
        # with no checksums — migrated transparently on first RMW
        # (ref compute-domain-kubelet-plugin/checkpoint_legacy.go).
        if "preparedClaims" in raw and not any(v in raw for v in self.SUPPORTED_VERSIONS):
            return from_dict(CheckpointData, raw, strict=False)
        for ver in reversed(self.SUPPORTED_VERSIONS):  # newest first
            entry = raw.get(ver)
            if entry is None:
                continue
            payload = entry.get("data")
            stored = entry.get("checksum")
            if payload is None or stored is None:
                raise CheckpointCorrupt(f"checkpoint {ver} entry missing data/checksum")
            actual = _checksum(payload)
            if actual != stored:
                diff = "\n".join(
                    difflib.unified_diff(
                        _canonical(payload).splitlines(),
                        [f"<stored checksum {stored} != computed {actual}>"],
                        lineterm="",
                    )
                )
                raise CheckpointCorrupt(
                    f"checkpoint {ver} checksum mismatch (stored {stored}, computed {actual}):\n"
                    f"{diff}"
                )
            return from_dict(CheckpointData, payload, strict=False)
        raise CheckpointCorrupt("checkpoint has no supported version payload")

    def _store_unlocked(self, data: CheckpointData, durable: bool = True) -> None:
        # Serialize ONCE, reusing per-claim fragments: the canonical payload
        # string both feeds the checksum and is spliced verbatim into each
        # version entry (payload JSON is canonical, so the checksum
        # validates on read).
        payload_s = data.canonical_payload()
        checksum = zlib.crc32(payload_s.encode("utf-8")) & 0xFFFFFFFF
        entry = '{"checksum":%d,"data":%s}' % (checksum, payload_s)
        raw_s = "{" + ",".join(f'"{v}":{entry}' for v in self.WRITE_VERSIONS) + "}"
        tmp = self.path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            f.write(raw_s)
            if durable:
                f.flush()
                # fdatasync: the claim payload must be durable before we
                # touch devices; the inode metadata (mtime) need not be.
                # os.replace below gives atomic visibility either way.
                os.fdatasync(f.fileno())
        os.replace(tmp, self.path)
        self._cache_payload = payload_s
        self._cache_stat = self._stat_key()

    # -- public API ---------------------------------------------------------

    def load(self, timeout: float = 10.0) -> CheckpointData:
        with self.lock.acquire(timeout=timeout):
            return self._load_unlocked()

    def update(self, mutate, timeout: float = 10.0, durable: bool = True) -> CheckpointData:
        """Locked read-modify-write (ref device_state.go:648-676).

        ``mutate(data)`` may return ``False`` to skip the write (read-only
        fast path). ``durable=False`` skips the fsync: safe only for
        mutations whose loss is recovered by an idempotent retry (e.g. claim
        removal after unprepare — replaying unprepare is a no-op).
        """
        with self.lock.acquire(timeout=timeout):
            if self._cache_obj is not None and self._stat_key() == self._cache_stat:
                data = self._cache_obj
            else:
                data = self._load_unlocked()
            try:
                if mutate(data) is False:
                    self._cache_obj = data
                    return data
            except BaseException:
                # the object may be partially mutated: drop it
                self._cache_obj = None
                raise
            self._store_unlocked(data, durable=durable)
            self._cache_obj = data
            return data
