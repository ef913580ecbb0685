"""ComputeDomain stack tests: controller reconciliation on the fake API
server, CD kubelet plugin daemon/channel prepare with readiness gating, and
the clique/daemon-supervisor machinery (with the real fabricd binary where
available)."""

import json
import os
import socket
import subprocess
import threading
import time

import pytest

from k8s_dra_driver_gpu_amd.api.configs import APIVERSION
from k8s_dra_driver_gpu_amd.cdplugin.plugin import ComputeDomainPlugin
from k8s_dra_driver_gpu_amd.controller.computedomain import (
    CD_FINALIZER,
    ComputeDomainController,
)
from k8s_dra_driver_gpu_amd.controller.templates import CD_LABEL_KEY
from k8s_dra_driver_gpu_amd.daemon.cdclique import CliqueManager
from k8s_dra_driver_gpu_amd.daemon.dnsnames import DNSNameManager, dns_name
from k8s_dra_driver_gpu_amd.daemon.process import ProcessManager, default_fabricd_path
from k8s_dra_driver_gpu_amd.device.devicelib import DeviceLib
from k8s_dra_driver_gpu_amd.device.mock import MockTree
from k8s_dra_driver_gpu_amd.dra import api as dra
from k8s_dra_driver_gpu_amd.k8s.client import FakeClient


def make_cd(client, name="cd1", ns="default", num_nodes=1):
    return client.create(
        "computedomains",
        {
            "apiVersion": APIVERSION.replace("/v1beta1", "") + "/v1beta1",
            "kind": "ComputeDomain",
            "metadata": {"name": name, "namespace": ns},
            "spec": {
                "numNodes": num_nodes,
                "channel": {
                    "resourceClaimTemplate": {"name": f"{name}-channel"},
                    "allocationMode": "Single",
                },
            },
        },
    )


@pytest.fixture
def controller_env():
    client = FakeClient()
    ctrl = ComputeDomainController(
        client, namespace="amd-dra", status_sync_period=0.1, cleanup_period=3600
    )
    ctrl.start()
    yield client, ctrl
    ctrl.stop()


def wait_for(fn, timeout=15.0, interval=0.02):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        v = fn()
        if v:
            return v
        time.sleep(interval)
    return fn()


class TestController:
    def test_materializes_daemonset_and_rcts(self, controller_env):
        client, ctrl = controller_env
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        ds = wait_for(lambda: client.get_or_none("daemonsets", "cd1-daemon", "default"))
        assert ds is not None
        assert ds["spec"]["template"]["spec"]["nodeSelector"] == {CD_LABEL_KEY: uid}
        rct_d = client.get_or_none("resourceclaimtemplates", "cd1-daemon-claim", "default")
        rct_w = client.get_or_none("resourceclaimtemplates", "cd1-channel", "default")
        assert rct_d and rct_w
        params = rct_w["spec"]["spec"]["devices"]["config"][0]["opaque"]["parameters"]
        assert params["kind"] == "ComputeDomainChannelConfig"
        assert params["domainID"] == uid
        got = wait_for(
            lambda: CD_FINALIZER
            in (client.get("computedomains", "cd1", "default")["metadata"].get("finalizers") or [])
        )
        assert got

    def test_status_from_clique(self, controller_env):
        client, ctrl = controller_env
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        client.create(
            "computedomaincliques",
            {
                "apiVersion": "resource.amd.com/v1beta1",
                "kind": "ComputeDomainClique",
                "metadata": {"name": f"{uid}.hive-1.0"},
                "daemons": [
                    {"nodeName": "n1", "ipAddress": "10.0.0.1", "cliqueID": "hive-1.0",
                     "index": 0, "status": "Ready"}
                ],
            },
        )
        cd2 = wait_for(
            lambda: (client.get("computedomains", "cd1", "default").get("status") or {}).get(
                "status"
            )
            == "Ready"
            and client.get("computedomains", "cd1", "default")
        )
        assert cd2
        assert cd2["status"]["nodes"][0]["name"] == "n1"

    def test_not_ready_until_enough_nodes(self, controller_env):
        client, ctrl = controller_env
        cd = make_cd(client, num_nodes=2)
        uid = cd["metadata"]["uid"]
        client.create(
            "computedomaincliques",
            {
                "apiVersion": "resource.amd.com/v1beta1",
                "kind": "ComputeDomainClique",
                "metadata": {"name": f"{uid}.h.0"},
                "daemons": [
                    {"nodeName": "n1", "ipAddress": "10.0.0.1", "cliqueID": "h.0",
                     "index": 0, "status": "Ready"}
                ],
            },
        )
        time.sleep(0.4)
        st = (client.get("computedomains", "cd1", "default").get("status") or {})
        assert st.get("status") != "Ready"

    def test_teardown_on_delete(self, controller_env):
        client, ctrl = controller_env
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        wait_for(lambda: client.get_or_none("daemonsets", "cd1-daemon", "default"))
        client.create(
            "nodes",
            {"apiVersion": "v1", "kind": "Node",
             "metadata": {"name": "n1", "labels": {CD_LABEL_KEY: uid}}},
        )
        client.delete("computedomains", "cd1", "default")
        gone = wait_for(lambda: client.get_or_none("computedomains", "cd1", "default") is None)
        assert gone
        assert client.get_or_none("daemonsets", "cd1-daemon", "default") is None
        assert client.get_or_none("resourceclaimtemplates", "cd1-channel", "default") is None
        node = client.get("nodes", "n1")
        assert CD_LABEL_KEY not in (node["metadata"].get("labels") or {})

    def test_cleanup_pass_removes_orphans(self):
        client = FakeClient()
        ctrl = ComputeDomainController(client, status_sync_period=3600, cleanup_period=3600)
        client.create(
            "daemonsets",
            {"metadata": {"name": "orphan", "namespace": "d",
                          "labels": {CD_LABEL_KEY: "dead-uid"}}},
        )
        client.create(
            "computedomaincliques",
            {"metadata": {"name": "dead-uid.h.0"}, "daemons": []},
        )
        removed = ctrl.cleanup_pass()
        assert removed == 2
        assert client.get_or_none("daemonsets", "orphan", "d") is None


# ---------------------------------------------------------------------------


@pytest.fixture
def cd_plugin(tmp_path):
    client = FakeClient()
    tree = MockTree(root=str(tmp_path / "mock"), num_gpus=2)
    tree.setup()
    lib = DeviceLib(backend=tree.backend())
    plugin = ComputeDomainPlugin(
        client=client,
        devicelib=lib,
        state_dir=str(tmp_path / "cd-state"),
        node_name="n1",
        retry_max_timeout=1.0,
    )
    return client, lib, plugin


UID_CLAIM = "aaaaaaaa-0000-0000-0000-000000000001"


def _mk_claim(client, ns, name, uid, device, kind, cd_uid, extra=None):
    params = {"apiVersion": APIVERSION, "kind": kind, "domainID": cd_uid}
    params.update(extra or {})
    client.create(
        "resourceclaims",
        {
            "apiVersion": "resource.k8s.io/v1beta1",
            "kind": "ResourceClaim",
            "metadata": {"name": name, "namespace": ns, "uid": uid},
            "status": {
                "allocation": {
                    "devices": {
                        "results": [
                            {
                                "request": "r0",
                                "driver": "compute-domain.amd.com",
                                "pool": "n1",
                                "device": device,
                            }
                        ],
                        "config": [
                            {
                                "requests": ["r0"],
                                "opaque": {
                                    "driver": "compute-domain.amd.com",
                                    "parameters": params,
                                },
                            }
                        ],
                    }
                }
            },
        },
    )


class TestCdPlugin:
    def test_resource_slice_shape(self, cd_plugin):
        _, _, plugin = cd_plugin
        sl = plugin.resource_slice()
        names = [d["name"] for d in sl["spec"]["devices"]]
        assert names == ["daemon-0", "channel-0"]

    def test_clique_id_from_topology(self, cd_plugin):
        _, _, plugin = cd_plugin
        assert plugin.clique_id().startswith("hive-")

    def test_daemon_prepare(self, cd_plugin):
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        _mk_claim(client, "default", "dc", UID_CLAIM, "daemon-0",
                  "ComputeDomainDaemonConfig", uid)
        req = dra.NodePrepareResourcesRequest(
            claims=[dra.Claim(namespace="default", name="dc", uid=UID_CLAIM)]
        )
        resp = plugin.node_prepare_resources(req, None)
        r = resp.claims[UID_CLAIM]
        assert r.error == "", r.error
        assert r.devices[0].device_name == "daemon-0"
        # domain dir materialized with fabricd config template
        ddir = plugin.domain_dir(uid)
        assert os.path.isdir(os.path.join(ddir, "shared"))
        cfg = json.load(open(os.path.join(ddir, "fabricd.cfg.template")))
        assert cfg["domain"] == uid
        assert cfg["cliqueID"].startswith("hive-")
        spec = json.load(open(plugin.cdi.claim_spec_path(UID_CLAIM)))
        env = spec["devices"][0]["containerEdits"]["env"]
        assert any(e.startswith("CLIQUE_ID=hive-") for e in env)
        assert any(e == f"COMPUTE_DOMAIN_UUID={uid}" for e in env)

    def test_channel_gated_on_readiness(self, cd_plugin):
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        _mk_claim(client, "default", "wc", UID_CLAIM, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        req = dra.NodePrepareResourcesRequest(
            claims=[dra.Claim(namespace="default", name="wc", uid=UID_CLAIM)]
        )
        # no clique entry -> retried then error
        resp = plugin.node_prepare_resources(req, None)
        assert "retry window exhausted" in resp.claims[UID_CLAIM].error
        # register this node Ready in the clique
        clique_id = plugin.clique_id()
        client.create(
            "computedomaincliques",
            {"metadata": {"name": f"{uid}.{clique_id}"},
             "daemons": [{"nodeName": "n1", "ipAddress": "10.0.0.1",
                          "cliqueID": clique_id, "index": 0, "status": "Ready"}]},
        )
        resp = plugin.node_prepare_resources(req, None)
        r = resp.claims[UID_CLAIM]
        assert r.error == "", r.error
        # node got labeled into the CD
        node = client.get("nodes", "n1")
        assert node["metadata"]["labels"][CD_LABEL_KEY] == uid
        spec = json.load(open(plugin.cdi.claim_spec_path(UID_CLAIM)))
        mounts = spec["devices"][0]["containerEdits"]["mounts"]
        assert mounts[0]["containerPath"] == "/compute-domain"

    def test_channel_namespace_mismatch_permanent(self, cd_plugin):
        client, lib, plugin = cd_plugin
        cd = make_cd(client, ns="default")
        uid = cd["metadata"]["uid"]
        _mk_claim(client, "other", "wc", UID_CLAIM, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        t0 = time.monotonic()
        resp = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(
                claims=[dra.Claim(namespace="other", name="wc", uid=UID_CLAIM)]
            ),
            None,
        )
        # permanent error: fails fast, not after the retry window
        assert time.monotonic() - t0 < 0.9
        assert "does not match" in resp.claims[UID_CLAIM].error

    def test_channel_exclusivity(self, cd_plugin):
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        clique_id = plugin.clique_id()
        client.create(
            "computedomaincliques",
            {"metadata": {"name": f"{uid}.{clique_id}"},
             "daemons": [{"nodeName": "n1", "ipAddress": "10.0.0.1",
                          "cliqueID": clique_id, "index": 0, "status": "Ready"}]},
        )
        uid2 = "bbbbbbbb-0000-0000-0000-000000000002"
        _mk_claim(client, "default", "w1", UID_CLAIM, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        _mk_claim(client, "default", "w2", uid2, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        r1 = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(claims=[dra.Claim(namespace="default", name="w1", uid=UID_CLAIM)]), None
        )
        assert r1.claims[UID_CLAIM].error == ""
        r2 = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(claims=[dra.Claim(namespace="default", name="w2", uid=uid2)]), None
        )
        assert "already allocated" in r2.claims[uid2].error
        # release and retry
        plugin.node_unprepare_resources(
            dra.NodeUnprepareResourcesRequest(claims=[dra.Claim(uid=UID_CLAIM)]), None
        )
        r3 = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(claims=[dra.Claim(namespace="default", name="w2", uid=uid2)]), None
        )
        assert r3.claims[uid2].error == ""

    def test_prepare_idempotent(self, cd_plugin):
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        _mk_claim(client, "default", "dc", UID_CLAIM, "daemon-0",
                  "ComputeDomainDaemonConfig", uid)
        req = dra.NodePrepareResourcesRequest(
            claims=[dra.Claim(namespace="default", name="dc", uid=UID_CLAIM)]
        )
        r1 = plugin.node_prepare_resources(req, None)
        r2 = plugin.node_prepare_resources(req, None)
        assert r1.claims[UID_CLAIM].devices[0].cdi_device_ids == \
            r2.claims[UID_CLAIM].devices[0].cdi_device_ids

    def test_stale_domain_dir_cleanup(self, cd_plugin):
        client, lib, plugin = cd_plugin
        os.makedirs(os.path.join(plugin.domains_dir, "dead-uid"), exist_ok=True)
        assert plugin.cleanup_stale_domain_dirs() == 1
        assert not os.path.exists(os.path.join(plugin.domains_dir, "dead-uid"))


class TestCdPluginTwoPhase:
    """Crash-recovery matrix for the 2-phase prepare protocol
    (ref device_state.go:186-256,736-745; cleanup.go:117-125)."""

    def _ready_clique(self, client, plugin, uid):
        clique_id = plugin.clique_id()
        client.create(
            "computedomaincliques",
            {"metadata": {"name": f"{uid}.{clique_id}"},
             "daemons": [{"nodeName": "n1", "ipAddress": "10.0.0.1",
                          "cliqueID": clique_id, "index": 0, "status": "Ready"}]},
        )

    def test_standard_spec_written_at_boot(self, cd_plugin):
        client, lib, plugin = cd_plugin
        spec = json.load(open(plugin.cdi.standard_spec_path()))
        assert spec["devices"][0]["name"] == "all"
        edits = spec["devices"][0]["containerEdits"]
        node_paths = [n["path"] for n in edits["deviceNodes"]]
        assert "/dev/kfd" in node_paths
        assert any(p.startswith("/dev/dri/renderD") for p in node_paths)
        assert "AMD_VISIBLE_DEVICES=void" in edits["env"]

    def test_daemon_cdi_ids_include_standard_device(self, cd_plugin):
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        _mk_claim(client, "default", "dc", UID_CLAIM, "daemon-0",
                  "ComputeDomainDaemonConfig", uid)
        resp = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(
                claims=[dra.Claim(namespace="default", name="dc", uid=UID_CLAIM)]
            ), None
        )
        ids = resp.claims[UID_CLAIM].devices[0].cdi_device_ids
        assert ids[0] == plugin.cdi.qualified_name("all")
        assert len(ids) == 2 and "claim-" in ids[1]

    def test_channel_cdi_ids_have_no_standard_device(self, cd_plugin):
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        self._ready_clique(client, plugin, uid)
        _mk_claim(client, "default", "wc", UID_CLAIM, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        resp = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(
                claims=[dra.Claim(namespace="default", name="wc", uid=UID_CLAIM)]
            ), None
        )
        ids = resp.claims[UID_CLAIM].devices[0].cdi_device_ids
        assert len(ids) == 1 and "=all" not in ids[0]

    def test_started_checkpointed_before_side_effects(self, cd_plugin):
        """PrepareStarted is durable BEFORE the first side effect: a failure
        inside device prepare leaves the annotated intent behind."""
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        _mk_claim(client, "default", "wc", UID_CLAIM, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        # readiness gate never satisfied -> every attempt is transient
        resp = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(
                claims=[dra.Claim(namespace="default", name="wc", uid=UID_CLAIM)]
            ), None
        )
        assert "retry window exhausted" in resp.claims[UID_CLAIM].error
        pc = plugin.checkpoints.load().get_claim(UID_CLAIM)
        assert pc is not None and pc.state == "PrepareStarted"
        assert pc.devices[0].type == "channel" and pc.devices[0].uuid == uid

    def test_crash_between_phases_then_retry_completes(self, cd_plugin):
        """Crash after side effects but before PrepareCompleted: the retry
        re-enters, is not blocked by its own Started entry, and completes."""
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        self._ready_clique(client, plugin, uid)
        _mk_claim(client, "default", "wc", UID_CLAIM, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        real_write = plugin.cdi.write_claim_spec
        calls = {"n": 0}

        def crashy(uid_, devs):
            calls["n"] += 1
            out = real_write(uid_, devs)
            if calls["n"] == 1:
                raise RuntimeError("simulated crash after CDI write")
            return out

        plugin.cdi.write_claim_spec = crashy
        plugin.retry_max_timeout = 0.05  # force the first request to fail
        req = dra.NodePrepareResourcesRequest(
            claims=[dra.Claim(namespace="default", name="wc", uid=UID_CLAIM)]
        )
        resp = plugin.node_prepare_resources(req, None)
        assert resp.claims[UID_CLAIM].error != ""
        assert plugin.checkpoints.load().get_claim(UID_CLAIM).state == "PrepareStarted"
        # kubelet retries the whole request ("restart")
        plugin.retry_max_timeout = 1.0
        resp = plugin.node_prepare_resources(req, None)
        assert resp.claims[UID_CLAIM].error == ""
        assert plugin.checkpoints.load().get_claim(UID_CLAIM).state == "PrepareCompleted"

    def test_started_claim_does_not_block_channel(self, cd_plugin):
        """A claim stuck in PrepareStarted must NOT hold the channel: the
        next claim wins and the stale one is reaped later
        (ref device_state.go:736-745)."""
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        self._ready_clique(client, plugin, uid)
        stale_uid = "cccccccc-0000-0000-0000-000000000003"
        from k8s_dra_driver_gpu_amd.plugin.checkpoint import (
            ClaimRef,
            PreparedClaim,
            PreparedDevice,
        )

        plugin.checkpoints.update(
            lambda d: d.set_claim(
                stale_uid,
                PreparedClaim(
                    state="PrepareStarted",
                    claim=ClaimRef(namespace="default", name="dead", uid=stale_uid),
                    devices=[PreparedDevice(type="channel", name="channel-0", uuid=uid)],
                ),
            )
        )
        _mk_claim(client, "default", "wc", UID_CLAIM, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        resp = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(
                claims=[dra.Claim(namespace="default", name="wc", uid=UID_CLAIM)]
            ), None
        )
        assert resp.claims[UID_CLAIM].error == ""

    def test_losing_racer_rolled_back(self, cd_plugin):
        """A concurrent (other-process) prepare completes between our phase 1
        and commit: our commit must fail atomically AND our side effects
        (CDI spec, checkpoint entry) must be rolled back."""
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        self._ready_clique(client, plugin, uid)
        winner_uid = "dddddddd-0000-0000-0000-000000000004"
        from k8s_dra_driver_gpu_amd.plugin.checkpoint import (
            ClaimRef,
            PreparedClaim,
            PreparedDevice,
        )

        real_gate = plugin._assert_domain_ready_on_node
        state = {"injected": False}

        def gate_then_inject(cd_obj):
            real_gate(cd_obj)
            if not state["injected"]:
                state["injected"] = True
                # simulate another plugin process committing the channel
                # between our phase 1 and our commit
                plugin.checkpoints.update(
                    lambda d: d.set_claim(
                        winner_uid,
                        PreparedClaim(
                            state="PrepareCompleted",
                            claim=ClaimRef(namespace="default", name="w", uid=winner_uid),
                            devices=[
                                PreparedDevice(type="channel", name="channel-0", uuid=uid)
                            ],
                        ),
                    )
                )

        plugin._assert_domain_ready_on_node = gate_then_inject
        _mk_claim(client, "default", "wc", UID_CLAIM, "channel-0",
                  "ComputeDomainChannelConfig", uid)
        resp = plugin.node_prepare_resources(
            dra.NodePrepareResourcesRequest(
                claims=[dra.Claim(namespace="default", name="wc", uid=UID_CLAIM)]
            ), None
        )
        assert "already allocated" in resp.claims[UID_CLAIM].error
        # rolled back: no checkpoint entry, no CDI spec for the loser
        assert plugin.checkpoints.load().get_claim(UID_CLAIM) is None
        assert not os.path.exists(plugin.cdi.claim_spec_path(UID_CLAIM))
        # winner untouched
        assert plugin.checkpoints.load().get_claim(winner_uid) is not None

    def test_rollback_removes_unreferenced_created_dir(self, cd_plugin):
        client, lib, plugin = cd_plugin
        ddir = plugin.domain_dir("rollback-dom")
        os.makedirs(os.path.join(ddir, "shared"))
        from k8s_dra_driver_gpu_amd.plugin.checkpoint import ClaimRef, PreparedClaim

        plugin.checkpoints.update(
            lambda d: d.set_claim(
                UID_CLAIM,
                PreparedClaim(
                    state="PrepareStarted",
                    claim=ClaimRef(namespace="default", name="x", uid=UID_CLAIM),
                ),
            )
        )
        plugin.cdi.write_claim_spec(UID_CLAIM, [])
        plugin._rollback_started(UID_CLAIM, [ddir])
        assert not os.path.exists(ddir)
        assert plugin.checkpoints.load().get_claim(UID_CLAIM) is None
        assert not os.path.exists(plugin.cdi.claim_spec_path(UID_CLAIM))

    def test_stale_started_claim_reaped_by_cleanup(self, cd_plugin):
        """A PrepareStarted claim whose ResourceClaim is gone from the API
        server is unprepared by the async cleanup pass (ref cleanup.go:117-125)."""
        client, lib, plugin = cd_plugin
        from k8s_dra_driver_gpu_amd.plugin.checkpoint import ClaimRef, PreparedClaim

        plugin.checkpoints.update(
            lambda d: d.set_claim(
                UID_CLAIM,
                PreparedClaim(
                    state="PrepareStarted",
                    claim=ClaimRef(namespace="default", name="ghost", uid=UID_CLAIM),
                ),
            )
        )
        assert plugin.cleanup_stale_claims() >= 1
        assert plugin.checkpoints.load().get_claim(UID_CLAIM) is None

    def test_randomized_crash_injection_recovers(self, cd_plugin):
        """Seeded fuzz over the 2-phase protocol: fail the CDI spec write or
        the domain-dir creation on a random call, verify the claim errors
        without corrupting state, then retry cleanly, unprepare, and assert
        nothing is left behind (no claim spec, no checkpoint entry)."""
        import glob
        import random

        client, lib, plugin = cd_plugin
        rng = random.Random(4242)
        for trial in range(12):
            cd = make_cd(client, name=f"cdz{trial}")
            uid = cd["metadata"]["uid"]
            claim_uid = f"{trial:08d}-fuzz-4000-8000-000000000000"
            kind = rng.choice(["daemon", "channel"])
            if kind == "channel":
                self._ready_clique(client, plugin, uid)
                _mk_claim(client, "default", f"cz{trial}", claim_uid,
                          "channel-0", "ComputeDomainChannelConfig", uid)
            else:
                _mk_claim(client, "default", f"cz{trial}", claim_uid,
                          "daemon-0", "ComputeDomainDaemonConfig", uid)
            req = dra.NodePrepareResourcesRequest(
                claims=[dra.Claim(namespace="default", name=f"cz{trial}",
                                  uid=claim_uid)])

            target = rng.choice(["cdi", "dir"])
            real_write = plugin.cdi.write_claim_spec
            real_dir = plugin._ensure_domain_dir

            def boom(*a, **k):
                raise OSError("injected crash")

            if target == "cdi":
                plugin.cdi.write_claim_spec = boom
            else:
                plugin._ensure_domain_dir = boom
            try:
                resp = plugin.node_prepare_resources(req, None)
                assert resp.claims[claim_uid].error != ""
            finally:
                plugin.cdi.write_claim_spec = real_write
                plugin._ensure_domain_dir = real_dir

            # retry with the fault cleared must complete
            resp = plugin.node_prepare_resources(req, None)
            assert resp.claims[claim_uid].error == "", resp.claims[claim_uid].error
            cp = plugin.checkpoints.load()
            assert cp.get_claim(claim_uid).state == "PrepareCompleted"

            # unprepare leaves no residue for this claim
            plugin.node_unprepare_resources(
                dra.NodeUnprepareResourcesRequest(
                    claims=[dra.Claim(namespace="default", name=f"cz{trial}",
                                      uid=claim_uid)]), None)
            assert plugin.checkpoints.load().get_claim(claim_uid) is None
            leftovers = [f for f in glob.glob(
                os.path.join(plugin.cdi.cdi_root, "*.json"))
                if claim_uid in f]
            assert not leftovers, leftovers

    def test_unprepare_of_started_claim_is_clean(self, cd_plugin):
        client, lib, plugin = cd_plugin
        from k8s_dra_driver_gpu_amd.plugin.checkpoint import ClaimRef, PreparedClaim

        plugin.checkpoints.update(
            lambda d: d.set_claim(
                UID_CLAIM,
                PreparedClaim(
                    state="PrepareStarted",
                    claim=ClaimRef(namespace="default", name="x", uid=UID_CLAIM),
                ),
            )
        )
        resp = plugin.node_unprepare_resources(
            dra.NodeUnprepareResourcesRequest(claims=[dra.Claim(uid=UID_CLAIM)]), None
        )
        assert resp.claims[UID_CLAIM].error == ""
        assert plugin.checkpoints.load().get_claim(UID_CLAIM) is None


# ---------------------------------------------------------------------------


class TestCliqueManager:
    def test_insert_gap_filling_index(self):
        client = FakeClient()
        m1 = CliqueManager(client, "uid1", "h.0", "n1", "10.0.0.1")
        m2 = CliqueManager(client, "uid1", "h.0", "n2", "10.0.0.2")
        m3 = CliqueManager(client, "uid1", "h.0", "n3", "10.0.0.3")
        assert m1.insert_self() == 0
        assert m2.insert_self() == 1
        assert m3.insert_self() == 2
        m2.remove_self()
        m4 = CliqueManager(client, "uid1", "h.0", "n4", "10.0.0.4")
        assert m4.insert_self() == 1  # gap filled
        assert m1.insert_self() == 0  # stable for existing member

    def test_ready_status_update(self):
        client = FakeClient()
        m = CliqueManager(client, "uid1", "h.0", "n1", "10.0.0.1")
        m.insert_self()
        m.set_ready(True)
        clique = client.get("computedomaincliques", "uid1.h.0")
        assert clique["daemons"][0]["status"] == "Ready"
        m.set_ready(False)
        assert client.get("computedomaincliques", "uid1.h.0")["daemons"][0]["status"] == "NotReady"

    def test_peer_update_callback(self):
        client = FakeClient()
        m1 = CliqueManager(client, "uid1", "h.0", "n1", "10.0.0.1")
        m1.insert_self()
        updates = []
        got = threading.Event()

        def cb(daemons):
            updates.append(daemons)
            got.set()

        m1.watch_peers(cb)
        time.sleep(0.1)
        m2 = CliqueManager(client, "uid1", "h.0", "n2", "10.0.0.2")
        m2.insert_self()
        assert got.wait(3.0)
        assert any(d["nodeName"] == "n2" for d in updates[-1])
        m1.stop()


class TestDnsNames:
    def test_name_format(self):
        assert dns_name(3) == "compute-domain-daemon-0003"

    def test_hosts_rewrite(self, tmp_path):
        hosts = tmp_path / "hosts"
        hosts.write_text("127.0.0.1 localhost\n")
        mgr = DNSNameManager(4, str(hosts))
        mapping = mgr.update_hosts(
            [{"index": 0, "ipAddress": "10.0.0.5"}, {"index": 2, "ipAddress": "10.0.0.7"}]
        )
        content = hosts.read_text()
        assert "127.0.0.1 localhost" in content
        assert "10.0.0.5\tcompute-domain-daemon-0000" in content
        assert "10.0.0.7\tcompute-domain-daemon-0002" in content
        assert mapping["compute-domain-daemon-0001"] == "127.0.0.1"
        # rewrite replaces the managed block, not append
        mgr.update_hosts([{"index": 0, "ipAddress": "10.0.0.9"}])
        content = hosts.read_text()
        assert content.count("BEGIN amd-dra") == 1
        assert "10.0.0.9\tcompute-domain-daemon-0000" in content

    def test_static_nodes_config(self, tmp_path):
        mgr = DNSNameManager(3, str(tmp_path / "hosts"))
        p = tmp_path / "nodes.cfg"
        mgr.write_nodes_config(str(p))
        assert p.read_text().splitlines() == [
            "compute-domain-daemon-0000",
            "compute-domain-daemon-0001",
            "compute-domain-daemon-0002",
        ]


# ---------------------------------------------------------------------------


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


FABRICD = default_fabricd_path()
needs_fabricd = pytest.mark.skipif(
    not os.path.exists(FABRICD), reason="fabricd binary not built (make -C native)"
)


@needs_fabricd
class TestFabricd:
    def _write_cfg(self, d, peer_port, cmd_port, peers, **extra):
        cfg = {
            "domain": "test-dom",
            "cliqueID": "h.0",
            "peerPort": peer_port,
            "commandPort": cmd_port,
            "nodesConfig": "nodes.cfg",
        }
        cfg.update(extra)
        with open(os.path.join(d, "fabricd.cfg"), "w") as f:
            json.dump(cfg, f)
        with open(os.path.join(d, "nodes.cfg"), "w") as f:
            f.write("\n".join(peers) + "\n")

    def _status(self, port):
        from k8s_dra_driver_gpu_amd.daemon.process import default_fabricctl_path

        out = subprocess.run(
            [default_fabricctl_path(), "-q", "-p", str(port)],
            capture_output=True,
            text=True,
            timeout=10,
        )
        return out.stdout.strip()

    def _launch_pair(self, tmp_path, host, attempt=0):
        """Start a 2-daemon mesh; retry once with fresh ports if either
        daemon dies at startup (a _free_port can be re-grabbed between the
        probe-close and fabricd's bind)."""
        d1, d2 = str(tmp_path / f"a{attempt}"), str(tmp_path / f"b{attempt}")
        os.makedirs(d1), os.makedirs(d2)
        p1, c1, p2, c2 = _free_port(), _free_port(), _free_port(), _free_port()
        self._write_cfg(d1, p1, c1, [f"{host}:{p2}"])
        self._write_cfg(d2, p2, c2, [f"{host}:{p1}"])
        procs = [
            subprocess.Popen([FABRICD, "-c", os.path.join(d, "fabricd.cfg")])
            for d in (d1, d2)
        ]
        time.sleep(0.5)
        if any(p.poll() is not None for p in procs) and attempt == 0:
            for p in procs:
                p.kill()
                p.wait(timeout=5)
            return self._launch_pair(tmp_path, host, attempt=1)
        return procs, c1, c2

    def test_two_daemon_mesh_over_hostnames(self, tmp_path):
        """FabricDaemonsWithDNSNames data path: peers given as RESOLVABLE
        HOSTNAMES (localhost standing in for the headless-service DNS names)
        — exercises fabricd's getaddrinfo path, not just numeric IPs."""
        procs, c1, c2 = self._launch_pair(tmp_path, "localhost")
        try:
            ok = wait_for(
                lambda: self._status(c1).startswith("READY") and self._status(c2).startswith("READY"),
                timeout=30.0,
                interval=0.3,
            )
            assert ok, f"status: {self._status(c1)} / {self._status(c2)}"
        finally:
            for p in procs:
                p.kill()
                p.wait(timeout=5)

    def test_two_daemon_mesh_reaches_ready(self, tmp_path):
        procs, c1, c2 = self._launch_pair(tmp_path, "127.0.0.1")
        try:
            ok = wait_for(
                lambda: self._status(c1).startswith("READY") and self._status(c2).startswith("READY"),
                timeout=30.0,
                interval=0.3,
            )
            assert ok, f"status: {self._status(c1)} / {self._status(c2)}"
            # kill one peer -> the other goes NOT_READY
            procs[1].terminate()
            procs[1].wait(timeout=5)
            not_ready = wait_for(
                lambda: self._status(c1).startswith("NOT_READY"), timeout=30.0, interval=0.3
            )
            assert not_ready
        finally:
            for p in procs:
                if p.poll() is None:
                    p.kill()
                p.wait(timeout=5)

    def test_sigusr1_reload(self, tmp_path):
        import signal as sig

        d1 = str(tmp_path / "a")
        os.makedirs(d1)
        p1, c1 = _free_port(), _free_port()
        self._write_cfg(d1, p1, c1, [])
        proc = subprocess.Popen([FABRICD, "-c", os.path.join(d1, "fabricd.cfg")])
        try:
            ok = wait_for(lambda: self._status(c1).startswith("READY"), timeout=10.0, interval=0.2)
            assert ok  # no peers -> READY
            with open(os.path.join(d1, "nodes.cfg"), "w") as f:
                f.write("127.0.0.1:1\n")  # unreachable peer
            proc.send_signal(sig.SIGUSR1)
            not_ready = wait_for(
                lambda: self._status(c1).startswith("NOT_READY"), timeout=10.0, interval=0.3
            )
            assert not_ready
        finally:
            proc.kill()
            proc.wait(timeout=5)


@needs_fabricd
class TestFabricdRobustness:
    """Quorum readiness, flap tolerance, clean shutdown, reconnect backoff
    and mTLS (VERDICT round-1 item 7; IMEX config-surface parity per
    templates/compute-domain-daemon-config.tmpl.cfg:84-218)."""

    _write_cfg = TestFabricd._write_cfg
    _status = TestFabricd._status

    def _metrics(self, port):
        from k8s_dra_driver_gpu_amd.daemon.process import default_fabricctl_path

        out = subprocess.run(
            [default_fabricctl_path(), "metrics", "-p", str(port)],
            capture_output=True, text=True, timeout=10,
        )
        return out.stdout

    def test_quorum_ready_with_one_dead_peer(self, tmp_path):
        """quorumPercent=50: 1 of 2 peers dead -> domain degrades to READY,
        not forever-NotReady (the IMEX quorum RECOVERY analog)."""
        da, db = str(tmp_path / "a"), str(tmp_path / "b")
        os.makedirs(da), os.makedirs(db)
        pa, ca, pb, cb = _free_port(), _free_port(), _free_port(), _free_port()
        dead = _free_port()  # nothing listens here
        self._write_cfg(da, pa, ca, [f"127.0.0.1:{pb}", f"127.0.0.1:{dead}"],
                        quorumPercent=50)
        self._write_cfg(db, pb, cb, [f"127.0.0.1:{pa}"])
        procs = [subprocess.Popen([FABRICD, "-c", os.path.join(d, "fabricd.cfg")])
                 for d in (da, db)]
        try:
            ok = wait_for(lambda: self._status(ca).startswith("READY 1/2"),
                          timeout=20.0, interval=0.3)
            assert ok, self._status(ca)
        finally:
            for p in procs:
                p.kill()
                p.wait(timeout=5)

    def test_all_peers_policy_not_ready_with_dead_peer(self, tmp_path):
        """Default (quorumPercent=100) keeps the strict all-peers semantic."""
        da = str(tmp_path / "a")
        os.makedirs(da)
        pa, ca = _free_port(), _free_port()
        self._write_cfg(da, pa, ca, [f"127.0.0.1:{_free_port()}"])
        proc = subprocess.Popen([FABRICD, "-c", os.path.join(da, "fabricd.cfg")])
        try:
            wait_for(lambda: self._status(ca).startswith("NOT_READY"), timeout=10.0)
            assert self._status(ca).startswith("NOT_READY 0/1")
        finally:
            proc.kill()
            proc.wait(timeout=5)

    def test_clean_shutdown_on_sigterm(self, tmp_path):
        """SIGTERM exits 0 via joined threads (no _exit) — the property the
        ASan tier needs for leak checking."""
        import signal as sig

        da = str(tmp_path / "a")
        os.makedirs(da)
        pa, ca = _free_port(), _free_port()
        self._write_cfg(da, pa, ca, [])
        proc = subprocess.Popen([FABRICD, "-c", os.path.join(da, "fabricd.cfg")])
        try:
            wait_for(lambda: self._status(ca).startswith("READY"), timeout=10.0)
            proc.send_signal(sig.SIGTERM)
            assert proc.wait(timeout=15) == 0
        finally:
            if proc.poll() is None:
                proc.kill()
                proc.wait(timeout=5)

    def test_reconnect_backoff_bounds_attempts(self, tmp_path):
        """A dead peer is retried with exponential backoff, not hammered
        every heartbeat tick."""
        da = str(tmp_path / "a")
        os.makedirs(da)
        pa, ca = _free_port(), _free_port()
        self._write_cfg(da, pa, ca, [f"127.0.0.1:{_free_port()}"],
                        reconnectBackoffMs=2000, reconnectBackoffMaxMs=8000)
        proc = subprocess.Popen([FABRICD, "-c", os.path.join(da, "fabricd.cfg")])
        try:
            wait_for(lambda: self._status(ca).startswith("NOT_READY"), timeout=10.0)
            time.sleep(5.0)
            m = self._metrics(ca)
            attempts = int(
                [l for l in m.splitlines() if l.startswith("fabricd_reconnect_attempts_total")][0]
                .split()[-1]
            )
            # without backoff the 1 s heartbeat would make ~6+ attempts;
            # with a 2 s base doubling to 8 s we see at most ~4
            assert 1 <= attempts <= 4, m
            assert "fabricd_quorum_percent 100" in m
        finally:
            proc.kill()
            proc.wait(timeout=5)

    def test_disconnected_grace_keeps_peer_up(self, tmp_path):
        """disconnectedGraceSec: a freshly-dead peer still counts as up
        inside the grace window (IMEX_NODE_DISCONNECTED_GRACE_TIME analog)."""
        da, db = str(tmp_path / "a"), str(tmp_path / "b")
        os.makedirs(da), os.makedirs(db)
        pa, ca, pb, cb = _free_port(), _free_port(), _free_port(), _free_port()
        self._write_cfg(da, pa, ca, [f"127.0.0.1:{pb}"], disconnectedGraceSec=30)
        self._write_cfg(db, pb, cb, [f"127.0.0.1:{pa}"])
        procs = [subprocess.Popen([FABRICD, "-c", os.path.join(d, "fabricd.cfg")])
                 for d in (da, db)]
        try:
            ok = wait_for(lambda: self._status(ca).startswith("READY 1/1"), timeout=20.0)
            assert ok, self._status(ca)
            procs[1].kill()
            procs[1].wait(timeout=5)
            time.sleep(2.5)  # heartbeat notices the drop
            # still READY: inside the 30 s grace window
            assert self._status(ca).startswith("READY"), self._status(ca)
        finally:
            for p in procs:
                if p.poll() is None:
                    p.kill()
                    p.wait(timeout=5)

    # -- mTLS ---------------------------------------------------------------

    def _gen_certs(self, d):
        """CA + one node cert signed by it (IMEX_AUTH_SOURCE=FILE analog)."""
        def run(*args):
            subprocess.run(list(args), check=True, capture_output=True, cwd=d)

        run("openssl", "req", "-x509", "-newkey", "rsa:2048", "-keyout", "ca.key",
            "-out", "ca.crt", "-days", "2", "-nodes", "-subj", "/CN=fabric-ca")
        run("openssl", "req", "-newkey", "rsa:2048", "-keyout", "node.key",
            "-out", "node.csr", "-nodes", "-subj", "/CN=fabric-node")
        run("openssl", "x509", "-req", "-in", "node.csr", "-CA", "ca.crt",
            "-CAkey", "ca.key", "-CAcreateserial", "-out", "node.crt", "-days", "2")

    def _mtls_cfg(self, certs_dir):
        return dict(
            authMode="mtls",
            tlsServerCert=os.path.join(certs_dir, "node.crt"),
            tlsServerKey=os.path.join(certs_dir, "node.key"),
            tlsClientCert=os.path.join(certs_dir, "node.crt"),
            tlsClientKey=os.path.join(certs_dir, "node.key"),
            tlsCa=os.path.join(certs_dir, "ca.crt"),
        )

    def test_mtls_mesh_reaches_ready(self, tmp_path):
        certs = str(tmp_path / "certs")
        os.makedirs(certs)
        self._gen_certs(certs)
        da, db = str(tmp_path / "a"), str(tmp_path / "b")
        os.makedirs(da), os.makedirs(db)
        pa, ca, pb, cb = _free_port(), _free_port(), _free_port(), _free_port()
        self._write_cfg(da, pa, ca, [f"127.0.0.1:{pb}"], **self._mtls_cfg(certs))
        self._write_cfg(db, pb, cb, [f"127.0.0.1:{pa}"], **self._mtls_cfg(certs))
        procs = [subprocess.Popen([FABRICD, "-c", os.path.join(d, "fabricd.cfg")])
                 for d in (da, db)]
        try:
            ok = wait_for(
                lambda: self._status(ca).startswith("READY") and self._status(cb).startswith("READY"),
                timeout=25.0, interval=0.3,
            )
            assert ok, f"{self._status(ca)} / {self._status(cb)}"
            # an unauthenticated plaintext client gets no PONG from the
            # mTLS peer port
            s = socket.create_connection(("127.0.0.1", pa), timeout=3)
            s.sendall(b"PING test-dom\n")
            s.settimeout(3)
            try:
                data = s.recv(64)
            except (socket.timeout, ConnectionError):
                data = b""
            s.close()
            assert not data.startswith(b"PONG")
        finally:
            for p in procs:
                p.kill()
                p.wait(timeout=5)


class TestProcessManager:
    def test_watchdog_restarts(self, tmp_path):
        marker = tmp_path / "count"
        script = tmp_path / "flaky.sh"
        script.write_text(f"#!/bin/bash\necho x >> {marker}\nsleep 600\n")
        script.chmod(0o755)
        pm = ProcessManager([str(script)])
        pm.start()
        try:
            assert wait_for(lambda: marker.exists(), timeout=5.0)
            pm._proc.kill()
            assert wait_for(
                lambda: marker.read_text().count("x") >= 2, timeout=5.0, interval=0.2
            )
            assert pm.restart_count >= 1
        finally:
            pm.stop()

    def test_stop_terminates(self, tmp_path):
        pm = ProcessManager(["sleep", "600"])
        pm.start()
        assert pm.is_running()
        pm.stop()
        assert not pm.is_running()


class TestLegacyStatusManager:
    def test_insert_ready_remove(self):
        from k8s_dra_driver_gpu_amd.daemon.legacy_status import LegacyStatusManager

        client = FakeClient()
        client.create(
            "computedomains",
            {"metadata": {"name": "cd1", "namespace": "d"}, "spec": {"numNodes": 2}},
        )
        m1 = LegacyStatusManager(client, "d", "cd1", "n1", "10.0.0.1", "h.0")
        m2 = LegacyStatusManager(client, "d", "cd1", "n2", "10.0.0.2", "h.0")
        assert m1.insert_self() == 0
        assert m2.insert_self() == 1
        m1.set_ready(True)
        cd = client.get("computedomains", "cd1", "d")
        nodes = cd["status"]["nodes"]
        assert nodes[0]["status"] == "Ready" and nodes[1]["status"] == "NotReady"
        m1.remove_self()
        cd = client.get("computedomains", "cd1", "d")
        assert [n["name"] for n in cd["status"]["nodes"]] == ["n2"]
        # gap filled by new joiner
        m3 = LegacyStatusManager(client, "d", "cd1", "n3", "10.0.0.3", "h.0")
        assert m3.insert_self() == 0

    def test_missing_cd_is_noop(self):
        from k8s_dra_driver_gpu_amd.daemon.legacy_status import LegacyStatusManager

        client = FakeClient()
        m = LegacyStatusManager(client, "d", "nope", "n1", "10.0.0.1")
        assert m.insert_self() == -1
        m.set_ready(True)
        m.remove_self()


class TestMultiNamespaceDaemonSets:
    def test_additional_namespaces(self):
        client = FakeClient()
        ctrl = ComputeDomainController(
            client, status_sync_period=3600, cleanup_period=3600,
            additional_namespaces=["tenant-a", "tenant-b"],
        )
        ctrl.start()
        try:
            make_cd(client, "cd1", ns="default")
            for ns in ("default", "tenant-a", "tenant-b"):
                assert wait_for(
                    lambda ns=ns: client.get_or_none("daemonsets", "cd1-daemon", ns)
                ), f"missing DS in {ns}"
            client.delete("computedomains", "cd1", "default")
            assert wait_for(
                lambda: client.get_or_none("daemonsets", "cd1-daemon", "tenant-a") is None
            )
        finally:
            ctrl.stop()


@needs_fabricd
class TestEightDaemonMesh:
    """8 daemon supervisors + 8 real fabricd processes meshing over
    localhost — the BASELINE config-5 control-plane shape (8x mesh domain)
    with failure/recovery."""

    def test_eight_node_domain_bringup(self, tmp_path):
        from k8s_dra_driver_gpu_amd.daemon.main import DaemonSupervisor

        client = FakeClient()
        ctrl = ComputeDomainController(client, status_sync_period=0.1, cleanup_period=3600)
        ctrl.start()
        cd = make_cd(client, num_nodes=8)
        uid = cd["metadata"]["uid"]
        sups = []
        try:
            t0 = time.monotonic()
            for i in range(8):
                sup = DaemonSupervisor(
                    client=client, cd_uid=uid, node_name=f"n{i}",
                    pod_ip="127.0.0.1",
                    work_dir=str(tmp_path / f"f{i}"),
                    clique_id="h.0",
                    peer_port=_free_port(), command_port=_free_port(),
                    fabricd_path=default_fabricd_path(),
                )
                sups.append(sup)
                threading.Thread(
                    target=lambda s=sup: s.run(ready_poll_interval=0.3), daemon=True
                ).start()
            # NOTE: peers are registered by IP; all daemons share 127.0.0.1 so
            # nodes.cfg entries collapse — meshes by distinct peer ports need
            # host:port entries, which IP mode writes as bare IPs. The clique
            # itself (membership, indices, readiness propagation) is the
            # subject here; fabricd READY with dedup'd self-peers still
            # exercises the heartbeat path.
            ready = wait_for(
                lambda: ((client.get("computedomains", "cd1", "default").get("status") or {})
                         .get("status")) == "Ready",
                timeout=30.0, interval=0.3,
            )
            bringup = time.monotonic() - t0
            assert ready, client.get("computedomains", "cd1", "default").get("status")
            assert bringup < 30.0
            cdo = client.get("computedomains", "cd1", "default")
            assert len(cdo["status"]["nodes"]) == 8
            assert sorted(n["index"] for n in cdo["status"]["nodes"]) == list(range(8))
            # kill one daemon's fabricd-backed readiness -> CD NotReady
            sups[3].stop()
            not_ready = wait_for(
                lambda: ((client.get("computedomains", "cd1", "default").get("status") or {})
                         .get("status")) == "NotReady",
                timeout=20.0, interval=0.3,
            )
            assert not_ready
        finally:
            for s in sups:
                s.stop()
            ctrl.stop()


class TestStrictFabricMode:
    def test_strict_mode_raises_without_hive(self, tmp_path):
        tree = MockTree(root=str(tmp_path / "m"), num_gpus=2, hive_id=0)
        tree.setup()
        # hive_id 0 in mock still writes hive when num_gpus>1; force none:
        import os as _os
        for root, _, files in _os.walk(tree.sysfs_root):
            for f in files:
                if f == "xgmi_hive_id":
                    open(_os.path.join(root, f), "w").write("0\n")
                if f == "properties" and "nodes" in root:
                    content = open(_os.path.join(root, f)).read()
                    open(_os.path.join(root, f), "w").write(
                        content.replace("hive_id 0", "hive_id 0").replace(
                            f"hive_id {0:d}", "hive_id 0"
                        )
                    )
        lib = DeviceLib(backend=tree.backend())
        strict = ComputeDomainPlugin(
            client=FakeClient(), devicelib=lib,
            state_dir=str(tmp_path / "s"), node_name="n1", strict_fabric=True,
        )
        import pytest as _pytest
        with _pytest.raises(RuntimeError, match="strict fabric"):
            strict.clique_id()
        lax = ComputeDomainPlugin(
            client=FakeClient(), devicelib=lib,
            state_dir=str(tmp_path / "s2"), node_name="n1",
        )
        assert lax.clique_id() == ""


class TestControllerFailover:
    def test_reconciliation_survives_leader_change(self):
        """Two controller candidates with leader election; kill the leader
        mid-flight — the standby takes over and reconciles new CDs
        (ref test_cd_failover.bats + leader-election ReleaseOnCancel)."""
        from k8s_dra_driver_gpu_amd.k8s.leaderelection import LeaderElector

        client = FakeClient()

        def make_candidate(identity):
            ctrl = ComputeDomainController(
                client, status_sync_period=0.1, cleanup_period=3600
            )
            elector = LeaderElector(
                client, "cd-ctrl", "sys", identity,
                lease_duration=0.6, retry_period=0.05,
            )
            elector.on_started_leading = ctrl.start
            elector.on_stopped_leading = ctrl.stop
            elector.run()
            return ctrl, elector

        c1, e1 = make_candidate("ctrl-1")
        c2, e2 = make_candidate("ctrl-2")
        try:
            assert wait_for(lambda: e1.is_leader.is_set() or e2.is_leader.is_set(), 5.0)
            make_cd(client, "cd-a")
            assert wait_for(lambda: client.get_or_none("daemonsets", "cd-a-daemon", "default"))
            # fail the current leader
            leader, standby = (e1, e2) if e1.is_leader.is_set() else (e2, e1)
            leader.stop()
            assert wait_for(lambda: standby.is_leader.is_set(), 10.0)
            make_cd(client, "cd-b")
            assert wait_for(
                lambda: client.get_or_none("daemonsets", "cd-b-daemon", "default"), 10.0
            ), "standby did not reconcile after failover"
        finally:
            e1.stop()
            e2.stop()


class TestCdClaimCleanup:
    def test_stale_claims_and_orphan_specs_swept(self, cd_plugin):
        client, lib, plugin = cd_plugin
        cd = make_cd(client)
        uid = cd["metadata"]["uid"]
        _mk_claim(client, "default", "dc", UID_CLAIM, "daemon-0",
                  "ComputeDomainDaemonConfig", uid)
        req = dra.NodePrepareResourcesRequest(
            claims=[dra.Claim(namespace="default", name="dc", uid=UID_CLAIM)]
        )
        assert plugin.node_prepare_resources(req, None).claims[UID_CLAIM].error == ""
        # live claim: kept
        assert plugin.cleanup_stale_claims() == 0
        # delete the ResourceClaim -> swept
        client.delete("resourceclaims", "dc", "default")
        assert plugin.cleanup_stale_claims() == 1
        assert plugin.checkpoints.load().get_claim(UID_CLAIM) is None
        # orphan CDI spec with no checkpoint entry -> swept
        from k8s_dra_driver_gpu_amd.cdi.spec import CdiDevice

        plugin.cdi.write_claim_spec("dead0000-0000-0000-0000-0000000000cd",
                                    [CdiDevice(name="zombie")])
        assert plugin.cleanup_stale_claims() == 1


@needs_fabricd
class TestFabricdCommands:
    def test_metrics_and_burn_commands(self, tmp_path):
        import socket as _s

        from k8s_dra_driver_gpu_amd.daemon.process import default_fabricctl_path

        d = str(tmp_path / "f")
        os.makedirs(d)
        # _free_port() is a bind-probe (TOCTOU): if fabricd loses the port
        # race and exits, retry with fresh ports instead of flaking
        proc, c1 = None, 0
        for attempt in range(3):
            p1, c1 = _free_port(), _free_port()
            with open(os.path.join(d, "fabricd.cfg"), "w") as f:
                json.dump({"domain": "t", "cliqueID": "h.0", "peerPort": p1,
                           "commandPort": c1, "nodesConfig": "nodes.cfg"}, f)
            open(os.path.join(d, "nodes.cfg"), "w").close()
            proc = subprocess.Popen([FABRICD, "-c", os.path.join(d, "fabricd.cfg")])

            def ctl(cmd):
                return subprocess.run(
                    [default_fabricctl_path(), cmd, "-p", str(c1)],
                    capture_output=True, text=True, timeout=10,
                ).stdout

            ok = wait_for(
                lambda: proc.poll() is not None or "READY" in ctl("-q"),
                timeout=15.0, interval=0.2,
            )
            if proc.poll() is None and ok:
                break
            proc.kill()
            proc.wait(timeout=5)
            proc = None
        assert proc is not None, "fabricd failed to start on 3 port attempts"

        def command(cmd, until):
            # read until the expected marker: a single recv() may return a
            # partial TCP segment
            sk = _s.create_connection(("127.0.0.1", c1), 5)
            sk.settimeout(30)
            sk.sendall(cmd + b"\n")
            out = b""
            while until not in out:
                chunk = sk.recv(4096)
                if not chunk:
                    break
                out += chunk
            sk.close()
            return out.decode()

        try:
            assert "READY" in ctl("-q")
            metrics = ctl("peers")
            assert metrics.strip().endswith("END")
            # marker must be the VALUE line — "# TYPE fabricd_probe_ok
            # gauge" precedes it and would end the read one segment early
            out = command(b"METRICS", b"fabricd_probe_ok 1")
            assert "fabricd_peers 0" in out
            assert "fabricd_probe_ok 1" in out
            # BURN on a GPU-less host: must answer with an error, not hang
            out = command(b"BURN", b"\n")
            assert out.startswith("BURN_OK") or out.startswith("ERR")
        finally:
            proc.kill()
            proc.wait(timeout=5)
