"""Full-stack e2e scenarios on the fake API server (the analog of the
reference's Ginkgo e2e + bats suites):

ResourceSlice publication -> DeviceClass/CEL selection by the scheduler stub
-> kubelet stub drives the plugin over real gRPC -> device mutation + CDI ->
unprepare; plus the full ComputeDomain bring-up with a live fabricd mesh.
"""

import json
import os
import socket
import time

import pytest
import yaml

from k8s_dra_driver_gpu_amd.cdi.spec import CdiHandler
from k8s_dra_driver_gpu_amd.cdplugin.plugin import ComputeDomainPlugin
from k8s_dra_driver_gpu_amd.controller.computedomain import ComputeDomainController
from k8s_dra_driver_gpu_amd.daemon.main import DaemonSupervisor
from k8s_dra_driver_gpu_amd.device.devicelib import DeviceLib
from k8s_dra_driver_gpu_amd.device.mock import MockTree
from k8s_dra_driver_gpu_amd.dra import api as dra
from k8s_dra_driver_gpu_amd.k8s.client import FakeClient
from k8s_dra_driver_gpu_amd.k8s.scheduler import SchedulerStub
from k8s_dra_driver_gpu_amd.plugin.checkpoint import CheckpointManager
from k8s_dra_driver_gpu_amd.plugin.device_state import DeviceState
from k8s_dra_driver_gpu_amd.plugin.driver import GpuDriver, k8s_claim_resolver
from k8s_dra_driver_gpu_amd.plugin.resourceslice import ResourceSliceGenerator

CHART = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "deployments", "helm", "amd-dra-driver",
)


def load_device_classes(client):
    from k8s_dra_driver_gpu_amd.utils.helmlite import render_chart

    rendered = render_chart(CHART, {"resourceApiVersion": "v1"})
    for doc in yaml.safe_load_all(rendered["deviceclasses.yaml"]):
        if doc:
            client.create("deviceclasses", doc)


def wait_for(fn, timeout=20.0, interval=0.05):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        v = fn()
        if v:
            return v
        time.sleep(interval)
    return fn()


@pytest.fixture
def stack(tmp_path):
    """GPU-plugin full stack on one fake node."""
    client = FakeClient()
    load_device_classes(client)
    tree = MockTree(root=str(tmp_path / "mock"), num_gpus=2)
    tree.setup()
    lib = DeviceLib(backend=tree.backend())
    state_dir = str(tmp_path / "state")
    cdi = CdiHandler(cdi_root=str(tmp_path / "cdi"), dev_root=tree.dev_root)
    ds = DeviceState(
        devicelib=lib, cdi=cdi,
        checkpoints=CheckpointManager(state_dir, boot_id="b1"), state_dir=state_dir,
    )
    driver = GpuDriver(state=ds, claim_resolver=k8s_claim_resolver(client), node_name="node-a")
    socks = driver.start(plugin_dir=str(tmp_path / "plugin"))
    kubelet = dra.DRAPluginClient(f"unix://{socks['dra']}")
    # publish slices (partitionable mode so counters are exercised)
    gen = ResourceSliceGenerator(lib, node_name="node-a", partitionable=True)
    for sl in gen.generate():
        client.apply("resourceslices", sl)
    sched = SchedulerStub(client)
    yield client, tree, lib, ds, driver, kubelet, sched, cdi
    kubelet.close()
    driver.stop()


def make_claim(client, name, ns="default", device_class="gpu.amd.com",
               selectors=None, configs=None):
    return client.create(
        "resourceclaims",
        {
            "apiVersion": "resource.k8s.io/v1beta1",
            "kind": "ResourceClaim",
            "metadata": {"name": name, "namespace": ns},
            "spec": {
                "devices": {
                    "requests": [
                        {
                            "name": "r0",
                            "deviceClassName": device_class,
                            **({"selectors": selectors} if selectors else {}),
                        }
                    ],
                    **({"config": configs} if configs else {}),
                }
            },
        },
    )


class TestGpuE2E:
    def test_slice_publication(self, stack):
        client, *_ = stack
        slices = client.list("resourceslices")
        assert len(slices) == 1
        names = [d["name"] for d in slices[0]["spec"]["devices"]]
        assert "gpu-0" in names and "gpu-0-cpx-7" in names

    def test_whole_gpu_lifecycle(self, stack):
        client, tree, lib, ds, driver, kubelet, sched, cdi = stack
        claim = make_claim(client, "c1")
        assert sched.schedule_pending() == 1
        claim = client.get("resourceclaims", "c1", "default")
        alloc = claim["status"]["allocation"]["devices"]["results"]
        assert alloc[0]["device"] in ("gpu-0", "gpu-8")
        uid = claim["metadata"]["uid"]
        resp = kubelet.prepare([dra.Claim(namespace="default", name="c1", uid=uid)])
        assert resp.claims[uid].error == ""
        cdi_id = resp.claims[uid].devices[0].cdi_device_ids[0]
        assert cdi_id.startswith("amd.com/gpu=")
        assert os.path.exists(cdi.claim_spec_path(uid))
        kubelet.unprepare([dra.Claim(namespace="default", name="c1", uid=uid)])
        assert not os.path.exists(cdi.claim_spec_path(uid))

    def test_cel_selection_by_product_name(self, stack):
        client, tree, lib, ds, driver, kubelet, sched, cdi = stack
        make_claim(
            client, "c-match",
            selectors=[{"cel": {"expression":
                'device.attributes["gpu.amd.com"].productName.matches("MI3[0-9]5X")'}}],
        )
        make_claim(
            client, "c-nomatch",
            selectors=[{"cel": {"expression":
                'device.attributes["gpu.amd.com"].productName.matches("H100")'}}],
        )
        sched.schedule_pending()
        assert (client.get("resourceclaims", "c-match", "default").get("status") or {}).get(
            "allocation"
        )
        assert not (client.get("resourceclaims", "c-nomatch", "default").get("status") or {}).get(
            "allocation"
        )

    def test_cel_selection_by_memory_capacity(self, stack):
        client, *_ , sched, cdi = stack
        make_claim(
            client, "c-mem",
            selectors=[{"cel": {"expression":
                'device.capacity["gpu.amd.com"].memory >= quantity("200Gi")'}}],
        )
        sched.schedule_pending()
        claim = client.get("resourceclaims", "c-mem", "default")
        dev = claim["status"]["allocation"]["devices"]["results"][0]["device"]
        assert dev in ("gpu-0", "gpu-8")  # whole GPUs satisfy 200Gi; partitions don't

    def test_eight_cpx_partitions_via_counters(self, stack):
        """BASELINE config 4 through the FULL stack: 8 claims, 8 CPX
        partitions of one GPU, counter math prevents a 9th."""
        client, tree, lib, ds, driver, kubelet, sched, cdi = stack
        for i in range(8):
            make_claim(
                client, f"p{i}", device_class="partition.gpu.amd.com",
                selectors=[{"cel": {"expression":
                    'device.attributes["gpu.amd.com"].computePartition == "CPX" && '
                    'device.name.matches("gpu-0-")'}}],
            )
        assert sched.schedule_pending() == 8
        devices = set()
        for i in range(8):
            claim = client.get("resourceclaims", f"p{i}", "default")
            uid = claim["metadata"]["uid"]
            dev = claim["status"]["allocation"]["devices"]["results"][0]["device"]
            devices.add(dev)
            resp = kubelet.prepare([dra.Claim(namespace="default", name=f"p{i}", uid=uid)])
            assert resp.claims[uid].error == "", resp.claims[uid].error
        assert devices == {f"gpu-0-cpx-{i}" for i in range(8)}
        assert lib.gpu_by_minor(0).compute_partition == "CPX"
        # whole gpu-0 now unallocatable (counters consumed)
        make_claim(client, "c-whole", selectors=[
            {"cel": {"expression": 'device.name == "gpu-0"'}}])
        assert sched.schedule_pending() == 0
        # unprepare all -> back to SPX
        for i in range(8):
            claim = client.get("resourceclaims", f"p{i}", "default")
            kubelet.unprepare([dra.Claim(namespace="default", name=f"p{i}",
                                         uid=claim["metadata"]["uid"])])
        assert lib.gpu_by_minor(0).compute_partition == "SPX"

    def test_whole_gpu_blocks_partitions(self, stack):
        """Reverse exclusivity (ref StaticMIG 'mutual exclusivity with
        physical GPU'): a held whole GPU consumes all its counters, so no
        partition of that GPU is allocatable until it is released."""
        client, tree, lib, ds, driver, kubelet, sched, cdi = stack
        make_claim(client, "whole0", selectors=[
            {"cel": {"expression": 'device.name == "gpu-0"'}}])
        assert sched.schedule_pending() == 1
        make_claim(
            client, "part0", device_class="partition.gpu.amd.com",
            selectors=[{"cel": {"expression":
                'device.name.matches("gpu-0-")'}}],
        )
        assert sched.schedule_pending() == 0  # blocked by the whole GPU
        # release the whole GPU -> the partition claim becomes allocatable
        whole = client.get("resourceclaims", "whole0", "default")
        sched.release(whole)
        client.delete("resourceclaims", "whole0", "default")
        assert sched.schedule_pending() == 1
        part = client.get("resourceclaims", "part0", "default")
        dev = part["status"]["allocation"]["devices"]["results"][0]["device"]
        assert dev.startswith("gpu-0-")

    def test_unschedulable_claim(self, stack):
        client, *_, sched, _ = stack
        make_claim(
            client, "c-none",
            selectors=[{"cel": {"expression":
                'device.attributes["gpu.amd.com"].architecture == "gfx1100"'}}],
        )
        assert sched.schedule_pending() == 0

    def test_tainted_device_skipped(self, stack):
        client, tree, lib, ds, driver, kubelet, sched, cdi = stack
        # republish with gpu-0 tainted
        gen = ResourceSliceGenerator(
            lib, node_name="node-a", partitionable=True,
            taints={"gpu-0": [{"key": "amd.com/gpu-unhealthy", "effect": "NoSchedule"}]},
        )
        for sl in gen.generate():
            client.apply("resourceslices", sl)
        make_claim(client, "c1", selectors=[
            {"cel": {"expression": 'device.attributes["gpu.amd.com"].type == "gpu"'}}])
        sched.schedule_pending()
        claim = client.get("resourceclaims", "c1", "default")
        dev = claim["status"]["allocation"]["devices"]["results"][0]["device"]
        assert dev == "gpu-8"  # healthy GPU picked


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


FABRICD = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "native", "bin", "fabricd"
)


@pytest.mark.skipif(not os.path.exists(FABRICD), reason="fabricd not built")
class TestComputeDomainE2E:
    def test_full_cd_bringup_and_channel(self, tmp_path):
        """BASELINE config 5 shape on CPU: CD created -> controller renders
        objects -> daemon supervisor joins clique, runs real fabricd, reports
        Ready -> controller mirrors status -> channel claim prepare succeeds."""
        client = FakeClient()
        load_device_classes(client)
        tree = MockTree(root=str(tmp_path / "mock"), num_gpus=2)
        tree.setup()
        lib = DeviceLib(backend=tree.backend())

        ctrl = ComputeDomainController(client, status_sync_period=0.1, cleanup_period=3600)
        ctrl.start()
        cd = client.create(
            "computedomains",
            {
                "apiVersion": "resource.amd.com/v1beta1",
                "kind": "ComputeDomain",
                "metadata": {"name": "cd1", "namespace": "default"},
                "spec": {"numNodes": 1,
                         "channel": {"resourceClaimTemplate": {"name": "cd1-channel"},
                                     "allocationMode": "Single"}},
            },
        )
        uid = cd["metadata"]["uid"]
        assert wait_for(lambda: client.get_or_none("daemonsets", "cd1-daemon", "default"))

        # "node-a" daemon pod: supervisor + real fabricd (no peers -> READY)
        t0 = time.monotonic()
        sup = DaemonSupervisor(
            client=client,
            cd_uid=uid,
            node_name="node-a",
            pod_ip="127.0.0.1",
            work_dir=str(tmp_path / "fabricd"),
            devicelib=lib,
            peer_port=_free_port(),
            command_port=_free_port(),
            fabricd_path=FABRICD,
        )
        import threading

        st = threading.Thread(target=lambda: sup.run(ready_poll_interval=0.2), daemon=True)
        st.start()
        try:
            cd_ready = wait_for(
                lambda: ((client.get("computedomains", "cd1", "default").get("status") or {})
                         .get("status")) == "Ready",
                timeout=20.0,
            )
            bringup_s = time.monotonic() - t0
            assert cd_ready, client.get("computedomains", "cd1", "default").get("status")
            assert bringup_s < 20.0  # well inside the reference's 20-min budget
            # channel claim through the CD plugin
            plugin = ComputeDomainPlugin(
                client=client, devicelib=lib, state_dir=str(tmp_path / "cd-state"),
                node_name="node-a", retry_max_timeout=5.0,
            )
            cuid = "cccccccc-0000-0000-0000-000000000001"
            client.create(
                "resourceclaims",
                {
                    "metadata": {"name": "wc", "namespace": "default", "uid": cuid},
                    "status": {"allocation": {"devices": {
                        "results": [{"request": "channel",
                                     "driver": "compute-domain.amd.com",
                                     "pool": "node-a", "device": "channel-0"}],
                        "config": [{"requests": ["channel"],
                                    "opaque": {"driver": "compute-domain.amd.com",
                                               "parameters": {
                                                   "apiVersion": "resource.amd.com/v1beta1",
                                                   "kind": "ComputeDomainChannelConfig",
                                                   "domainID": uid}}}],
                    }}},
                },
            )
            resp = plugin.node_prepare_resources(
                dra.NodePrepareResourcesRequest(
                    claims=[dra.Claim(namespace="default", name="wc", uid=cuid)]
                ),
                None,
            )
            assert resp.claims[cuid].error == "", resp.claims[cuid].error
            # membership snapshot published for RCCL bootstrap
            members = json.load(
                open(os.path.join(str(tmp_path / "fabricd"), "shared", "members.json"))
            )
            assert members["domain"] == uid
            assert members["daemons"][0]["nodeName"] == "node-a"
        finally:
            sup.stop()
            ctrl.stop()

    def test_cd_deletion_cleans_clique(self, tmp_path):
        client = FakeClient()
        ctrl = ComputeDomainController(client, status_sync_period=0.1, cleanup_period=3600)
        ctrl.start()
        cd = client.create(
            "computedomains",
            {"metadata": {"name": "cd1", "namespace": "default"},
             "spec": {"numNodes": 1}},
        )
        uid = cd["metadata"]["uid"]
        wait_for(lambda: client.get_or_none("daemonsets", "cd1-daemon", "default"))
        client.create("computedomaincliques",
                      {"metadata": {"name": f"{uid}.h.0"}, "daemons": []})
        client.delete("computedomains", "cd1", "default")
        assert wait_for(lambda: client.get_or_none("computedomains", "cd1", "default") is None)
        assert wait_for(lambda: client.get_or_none("computedomaincliques", f"{uid}.h.0") is None)
        ctrl.stop()


@pytest.mark.skipif(not os.path.exists(FABRICD), reason="fabricd not built")
class TestLocalClusterDemo:
    """The demo specs through the LocalCluster harness (bats-suite analog)."""

    def test_quickstart_specs(self, tmp_path):
        from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

        specs = os.path.join(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            "demo", "specs", "quickstart",
        )
        cluster = LocalCluster(num_gpus=16, work_dir=str(tmp_path)).start()
        try:
            ev1 = cluster.apply_yaml(os.path.join(specs, "gpu-test1.yaml"))
            assert any("prepared gpu-" in e for e in ev1), ev1
            ev3 = cluster.apply_yaml(os.path.join(specs, "gpu-test-partitions.yaml"))
            prepared = [e for e in ev3 if "prepared gpu-" in e and "-cpx-" in e]
            assert len(prepared) == 8, ev3
            ev6 = cluster.apply_yaml(os.path.join(specs, "gpu-test3.yaml"))
            assert len([e for e in ev6 if "prepared gpu-" in e]) == 2, ev6
            ev5 = cluster.apply_yaml(os.path.join(specs, "gpu-test-extres.yaml"))
            assert any("(extended-resource)" in e for e in ev5), ev5
            ev2 = cluster.apply_yaml(os.path.join(specs, "gpu-test2.yaml"))
            assert any("prepared gpu-" in e for e in ev2), ev2
            ev7 = cluster.apply_yaml(os.path.join(specs, "gpu-test5.yaml"))
            assert any("prepared gpu-" in e for e in ev7), ev7
            # two distinct GPUs in one pod (two claims)
            ev8 = cluster.apply_yaml(os.path.join(specs, "gpu-test7.yaml"))
            got = [e for e in ev8 if "prepared gpu-" in e]
            assert len(got) == 2 and len(set(got)) == 2, ev8
            # one claim asking two devices (count: 2)
            ev9 = cluster.apply_yaml(os.path.join(specs, "gpu-test8.yaml"))
            two = [e for e in ev9 if "prepared gpu-" in e]
            assert len(two) >= 1, ev9
            ev4 = cluster.apply_yaml(os.path.join(specs, "cd-test1.yaml"))
            assert cluster.wait_cd_ready("cd1", "cd-test1"), ev4
            assert any("prepared channel-0" in e for e in ev4), ev4
            # pod deletion unprepares and frees the devices
            cluster.delete_pod("gpu-test1", "pod1")
        finally:
            cluster.stop()

    def test_quickstart_vfio_spec(self, tmp_path):
        # gpu-test-vfio end-to-end on the mock vfio manager: the claim
        # rebinds the device and the CDI spec injects the vfio group node
        from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

        specs = os.path.join(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            "demo", "specs", "quickstart",
        )
        cluster = LocalCluster(num_gpus=2, work_dir=str(tmp_path),
                               vfio=True).start()
        try:
            ev = cluster.apply_yaml(os.path.join(specs, "gpu-test-vfio.yaml"))
            assert any("prepared" in e for e in ev), ev
            import glob as _glob
            import json as _json
            found = []
            for f in _glob.glob(os.path.join(str(tmp_path), "cdi", "*claim*.json")):
                spec = _json.load(open(f))
                for d in spec["devices"]:
                    found += [n["path"] for n in
                              d.get("containerEdits", {}).get("deviceNodes", []) or []]
            assert any("/dev/vfio/" in p for p in found), found
        finally:
            cluster.stop()


@pytest.mark.skipif(not os.path.exists(FABRICD), reason="fabricd not built")
class TestSharedClaim:
    """N pods share one ResourceClaim (the reference e2e
    'N-pods-share-one-claim' scenario, gpu_allocation_test.go:86-228)."""

    def test_two_pods_one_claim(self, tmp_path):
        from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

        cluster = LocalCluster(num_gpus=1, work_dir=str(tmp_path)).start()
        try:
            cluster.client.create(
                "resourceclaims",
                {
                    "apiVersion": "resource.k8s.io/v1beta1",
                    "kind": "ResourceClaim",
                    "metadata": {"name": "shared", "namespace": "default"},
                    "spec": {"devices": {"requests": [
                        {"name": "gpu", "deviceClassName": "gpu.amd.com"}]}},
                },
            )
            for pod in ("pod-a", "pod-b"):
                ev = cluster._run_workload(
                    {
                        "apiVersion": "v1",
                        "kind": "Pod",
                        "metadata": {"name": pod, "namespace": "default"},
                        "spec": {
                            "containers": [{"name": "c", "resources": {"claims": [{"name": "g"}]}}],
                            "resourceClaims": [{"name": "g", "resourceClaimName": "shared"}],
                        },
                    }
                )
                assert any("(shared)" in e for e in ev), ev
            # both pods saw the SAME device via the idempotent prepare
            claim = cluster.client.get("resourceclaims", "shared", "default")
            uid = claim["metadata"]["uid"]
            assert cluster._prepared_pods["default/pod-a"] == [f"gpu.amd.com:{uid}"]
            assert cluster._prepared_pods["default/pod-b"] == [f"gpu.amd.com:{uid}"]
        finally:
            cluster.stop()


class TestSchedulerRestart:
    def test_restarted_scheduler_respects_existing_allocations(self, stack):
        """A new SchedulerStub instance (scheduler restart) rebuilds its
        bookkeeping from claim statuses and refuses to double-allocate."""
        client, tree, lib, ds, driver, kubelet, sched, cdi = stack
        make_claim(client, "held", selectors=[
            {"cel": {"expression": 'device.name == "gpu-0"'}}])
        assert sched.schedule_pending() == 1
        # fresh scheduler (simulating restart)
        sched2 = SchedulerStub(client)
        make_claim(client, "wants-same", selectors=[
            {"cel": {"expression": 'device.name == "gpu-0"'}}])
        assert sched2.schedule_pending() == 0  # gpu-0 already held
        make_claim(client, "other")
        assert sched2.schedule_pending() == 1  # gpu-8 still free
        claim = client.get("resourceclaims", "other", "default")
        assert claim["status"]["allocation"]["devices"]["results"][0]["device"] == "gpu-8"

    def test_restart_respects_partition_counters(self, stack):
        client, tree, lib, ds, driver, kubelet, sched, cdi = stack
        for i in range(8):
            make_claim(client, f"p{i}", device_class="partition.gpu.amd.com",
                       selectors=[{"cel": {"expression":
                           'device.name.matches("gpu-0-cpx-")'}}])
        assert sched.schedule_pending() == 8
        sched2 = SchedulerStub(client)
        make_claim(client, "whole0", selectors=[
            {"cel": {"expression": 'device.name == "gpu-0"'}}])
        assert sched2.schedule_pending() == 0  # counters consumed by partitions

@pytest.mark.skipif(not os.path.exists(FABRICD), reason="fabricd not built")
class TestExtendedResource:
    """Legacy `amd.com/gpu: N` container limits satisfied through DRA —
    the DRAExtendedResource path (k8s >= 1.35) the reference enables via
    extendedResourceName on its GPU DeviceClass (deviceclass-gpu.yaml:13;
    scenario tests/bats/test_gpu_extres.bats)."""

    def test_legacy_gpu_limit_via_dra(self, tmp_path):
        from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

        cluster = LocalCluster(num_gpus=2, work_dir=str(tmp_path)).start()
        try:
            ev = cluster._run_workload({
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": "legacy", "namespace": "default"},
                "spec": {"containers": [{"name": "c", "resources": {
                    "limits": {"amd.com/gpu": 2}}}]},
            })
            assert any("(extended-resource)" in e for e in ev), ev
            pod = cluster.client.get("pods", "legacy", "default")
            ercs = pod["status"]["extendedResourceClaimStatus"]
            assert ercs["resourceClaimName"] == "legacy-extended-resources"
            assert ercs["requestMappings"][0]["resourceName"] == "amd.com/gpu"
            claim = cluster.client.get(
                "resourceclaims", "legacy-extended-resources", "default")
            results = claim["status"]["allocation"]["devices"]["results"]
            assert len(results) == 2
            assert len({r["device"] for r in results}) == 2  # distinct GPUs
            cluster.delete_pod("default", "legacy")
        finally:
            cluster.stop()

    def test_oversized_request_stays_pending(self, tmp_path):
        from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

        cluster = LocalCluster(num_gpus=1, work_dir=str(tmp_path)).start()
        try:
            ev = cluster._run_workload({
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": "greedy", "namespace": "default"},
                "spec": {"containers": [{"name": "c", "resources": {
                    "limits": {"amd.com/gpu": 3}}}]},
            })
            assert not any("prepared" in e for e in ev), ev
            pod = cluster.client.get("pods", "greedy", "default")
            assert not (pod.get("status") or {}).get("extendedResourceClaimStatus")
        finally:
            cluster.stop()

    def test_unmapped_resource_ignored(self, tmp_path):
        from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

        cluster = LocalCluster(num_gpus=1, work_dir=str(tmp_path)).start()
        try:
            ev = cluster._run_workload({
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": "cpuonly", "namespace": "default"},
                "spec": {"containers": [{"name": "c", "resources": {
                    "limits": {"cpu": 2, "memory": "1Gi"}}}]},
            })
            assert not any("prepared" in e for e in ev), ev
        finally:
            cluster.stop()

@pytest.mark.skipif(not os.path.exists(FABRICD), reason="fabricd not built")
class TestTwoTemplatesTwoGpus:
    """Pod with two ResourceClaimTemplates gets two DISTINCT GPUs (ref
    scenario tests/bats/test_gpu_robustness.bats:117)."""

    def test_pod_with_two_rcts(self, tmp_path):
        from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

        cluster = LocalCluster(num_gpus=2, work_dir=str(tmp_path)).start()
        try:
            for t in ("rct-a", "rct-b"):
                cluster.client.create("resourceclaimtemplates", {
                    "apiVersion": "resource.k8s.io/v1beta1",
                    "kind": "ResourceClaimTemplate",
                    "metadata": {"name": t, "namespace": "default"},
                    "spec": {"spec": {"devices": {"requests": [
                        {"name": "gpu", "deviceClassName": "gpu.amd.com"}]}}},
                })
            ev = cluster._run_workload({
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": "dual", "namespace": "default"},
                "spec": {
                    "containers": [{"name": "c", "resources": {
                        "claims": [{"name": "g0"}, {"name": "g1"}]}}],
                    "resourceClaims": [
                        {"name": "g0", "resourceClaimTemplateName": "rct-a"},
                        {"name": "g1", "resourceClaimTemplateName": "rct-b"},
                    ],
                },
            })
            prepared = [e for e in ev if "prepared gpu-" in e]
            assert len(prepared) == 2, ev
            devs = set()
            for name in ("dual-g0", "dual-g1"):
                claim = cluster.client.get("resourceclaims", name, "default")
                devs.add(claim["status"]["allocation"]["devices"]["results"][0]["device"])
            assert len(devs) == 2, devs  # distinct GPUs
            cluster.delete_pod("default", "dual")
        finally:
            cluster.stop()

@pytest.mark.skipif(not os.path.exists(FABRICD), reason="fabricd not built")
class TestReacquireAfterDelete:
    """GPU allocated on pod create, released on pod delete, re-acquired by
    the next pod (ref tests/bats/test_gpu_robustness.bats:57+85)."""

    def test_single_gpu_reacquired(self, tmp_path):
        from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

        cluster = LocalCluster(num_gpus=1, work_dir=str(tmp_path)).start()
        try:
            cluster.client.create("resourceclaimtemplates", {
                "apiVersion": "resource.k8s.io/v1beta1",
                "kind": "ResourceClaimTemplate",
                "metadata": {"name": "one-gpu", "namespace": "default"},
                "spec": {"spec": {"devices": {"requests": [
                    {"name": "gpu", "deviceClassName": "gpu.amd.com"}]}}},
            })

            def pod(name):
                return {
                    "apiVersion": "v1", "kind": "Pod",
                    "metadata": {"name": name, "namespace": "default"},
                    "spec": {
                        "containers": [{"name": "c", "resources": {
                            "claims": [{"name": "g"}]}}],
                        "resourceClaims": [
                            {"name": "g", "resourceClaimTemplateName": "one-gpu"}],
                    },
                }

            cluster.schedule_timeout = 1.0
            ev1 = cluster._run_workload(pod("first"))
            assert any("prepared gpu-" in e for e in ev1), ev1
            # the only GPU is held: a second pod is unschedulable
            ev2 = cluster._run_workload(pod("second"))
            assert any("unschedulable" in e for e in ev2), ev2
            # delete the holder: its claim is GC'd and the device released —
            # the waiting second pod's pending claim acquires it
            cluster.delete_pod("default", "first")
            assert cluster.client.get_or_none(
                "resourceclaims", "first-g", "default") is None
            cluster.scheduler.schedule_pending()
            second = cluster.client.get("resourceclaims", "second-g", "default")
            assert (second.get("status") or {}).get("allocation"), second
            # and once the second pod is gone too, a brand-new pod re-acquires
            cluster.delete_pod("default", "second")
            ev3 = cluster._run_workload(pod("third"))
            assert any("prepared gpu-" in e for e in ev3), ev3
        finally:
            cluster.stop()
