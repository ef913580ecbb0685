"""Tests for device health -> taints, the healthcheck service, checkpoint
cleanup, and VFIO passthrough (mocked sysfs)."""

import os
import threading

import pytest

from k8s_dra_driver_gpu_amd.cdi.spec import CdiHandler
from k8s_dra_driver_gpu_amd.device.devicelib import DeviceLib
from k8s_dra_driver_gpu_amd.device.mock import MockTree
from k8s_dra_driver_gpu_amd.k8s.client import FakeClient
from k8s_dra_driver_gpu_amd.plugin.checkpoint import CheckpointManager, ClaimRef
from k8s_dra_driver_gpu_amd.plugin.cleanup import CheckpointCleanupManager
from k8s_dra_driver_gpu_amd.plugin.device_health import (
    EVENT_ECC_CORRECTABLE,
    EVENT_ECC_UNCORRECTABLE,
    HealthEvent,
    HealthMonitor,
    SysfsRasSource,
    TaintTracker,
)
from k8s_dra_driver_gpu_amd.plugin.device_state import (
    AllocatedClaim,
    AllocatedDevice,
    DeviceState,
)
from k8s_dra_driver_gpu_amd.plugin.driver import GpuDriver, static_claim_resolver
from k8s_dra_driver_gpu_amd.plugin.health_svc import HealthServer, check_health
from k8s_dra_driver_gpu_amd.plugin.vfio import VfioError, VfioPciManager

UID1 = "11111111-1111-1111-1111-111111111111"


@pytest.fixture
def env(tmp_path):
    tree = MockTree(root=str(tmp_path / "mock"), num_gpus=2)
    tree.setup()
    lib = DeviceLib(backend=tree.backend())
    return tree, lib


def _write_ras(tree, gpu_minor, ue=0, ce=0):
    dev = os.path.join(tree.sysfs_root, "class", "drm", f"card{gpu_minor}", "device", "ras")
    os.makedirs(dev, exist_ok=True)
    with open(os.path.join(dev, "ue_count"), "w") as f:
        f.write(str(ue))
    with open(os.path.join(dev, "ce_count"), "w") as f:
        f.write(str(ce))


class TestHealthMonitor:
    def test_ras_counter_events(self, env):
        tree, lib = env
        _write_ras(tree, 0, ue=0, ce=0)
        src = SysfsRasSource(lib)
        assert src.poll() == []  # baseline capture
        _write_ras(tree, 0, ue=2, ce=1)
        events = src.poll()
        kinds = {e.kind for e in events}
        assert EVENT_ECC_UNCORRECTABLE in kinds
        assert EVENT_ECC_CORRECTABLE in kinds
        assert src.poll() == []  # no further increase

    def test_skip_list_classification(self, env):
        tree, lib = env
        got = []
        mon = HealthMonitor(lib, got.extend, use_amdsmi=False, poll_interval=3600)
        events = mon.classify(
            [
                HealthEvent("u1", EVENT_ECC_CORRECTABLE),
                HealthEvent("u1", EVENT_ECC_UNCORRECTABLE),
            ]
        )
        assert events[0].fatal is False
        assert events[1].fatal is True

    def test_additional_skip(self, env):
        tree, lib = env
        mon = HealthMonitor(
            lib, lambda e: None, additional_skip={EVENT_ECC_UNCORRECTABLE},
            use_amdsmi=False, poll_interval=3600,
        )
        evs = mon.classify([HealthEvent("u1", EVENT_ECC_UNCORRECTABLE)])
        assert evs[0].fatal is False

    def test_taint_tracker_republish(self, env):
        tree, lib = env
        published = []
        tracker = TaintTracker(lib, published.append)
        g0 = lib.gpus()[0]
        tracker.on_events([HealthEvent(g0.uuid, EVENT_ECC_UNCORRECTABLE, fatal=True)])
        assert len(published) == 1
        taints = published[0][g0.canonical_name]
        assert taints[0]["key"] == "amd.com/gpu-unhealthy"
        assert taints[0]["effect"] == "NoSchedule"
        # non-fatal events do not republish
        tracker.on_events([HealthEvent(g0.uuid, EVENT_ECC_CORRECTABLE, fatal=False)])
        assert len(published) == 1
        # duplicate taint does not republish
        tracker.on_events([HealthEvent(g0.uuid, EVENT_ECC_UNCORRECTABLE, fatal=True)])
        assert len(published) == 1

    def test_end_to_end_taint_in_slice(self, env, tmp_path):
        from k8s_dra_driver_gpu_amd.plugin.resourceslice import ResourceSliceGenerator

        tree, lib = env
        _write_ras(tree, 0, ue=0)
        slices = []

        def republish(taints):
            gen = ResourceSliceGenerator(lib, node_name="n1", taints=taints)
            slices.append(gen.generate()[0])

        tracker = TaintTracker(lib, republish)
        mon = HealthMonitor(lib, tracker.on_events, use_amdsmi=False, poll_interval=3600)
        mon.poll_once()  # baseline
        _write_ras(tree, 0, ue=1)
        mon.poll_once()
        assert slices, "fatal event should republish"
        dev = next(d for d in slices[-1]["spec"]["devices"] if d["name"] == "gpu-0")
        assert dev["basic"]["taints"][0]["key"] == "amd.com/gpu-unhealthy"


class TestHealthService:
    def test_serving_when_sockets_ok(self, env, tmp_path):
        tree, lib = env
        ds = DeviceState(
            devicelib=lib,
            cdi=CdiHandler(cdi_root=str(tmp_path / "cdi"), dev_root=tree.dev_root),
            checkpoints=CheckpointManager(str(tmp_path / "state")),
            state_dir=str(tmp_path / "state"),
        )
        driver = GpuDriver(state=ds, claim_resolver=static_claim_resolver({}), node_name="n1")
        socks = driver.start(plugin_dir=str(tmp_path / "p"), registry_dir=str(tmp_path / "r"))
        hs = HealthServer(socks["dra"], socks["registration"])
        port = hs.start()
        try:
            assert check_health(port) is True
            driver.stop(grace=0.1)
            assert check_health(port) is False
        finally:
            hs.stop()


class TestCheckpointCleanup:
    def _state(self, tmp_path, lib, tree):
        return DeviceState(
            devicelib=lib,
            cdi=CdiHandler(cdi_root=str(tmp_path / "cdi"), dev_root=tree.dev_root),
            checkpoints=CheckpointManager(str(tmp_path / "state")),
            state_dir=str(tmp_path / "state"),
        )

    def test_removes_orphaned_claims(self, env, tmp_path):
        tree, lib = env
        ds = self._state(tmp_path, lib, tree)
        client = FakeClient()
        ds.prepare(
            AllocatedClaim(
                ref=ClaimRef(namespace="d", name="c1", uid=UID1),
                devices=[AllocatedDevice(device="gpu-0")],
            )
        )
        mgr = CheckpointCleanupManager(ds, client, interval=3600)
        assert mgr.cleanup_pass() == 1
        assert ds.prepared_claims() == {}

    def test_keeps_live_claims(self, env, tmp_path):
        tree, lib = env
        ds = self._state(tmp_path, lib, tree)
        client = FakeClient()
        client.create(
            "resourceclaims",
            {"metadata": {"name": "c1", "namespace": "d", "uid": UID1}},
        )
        ds.prepare(
            AllocatedClaim(
                ref=ClaimRef(namespace="d", name="c1", uid=UID1),
                devices=[AllocatedDevice(device="gpu-0")],
            )
        )
        mgr = CheckpointCleanupManager(ds, client, interval=3600)
        assert mgr.cleanup_pass() == 0
        assert UID1 in ds.prepared_claims()

    def test_uid_mismatch_cleaned(self, env, tmp_path):
        tree, lib = env
        ds = self._state(tmp_path, lib, tree)
        client = FakeClient()
        client.create(
            "resourceclaims",
            {"metadata": {"name": "c1", "namespace": "d",
                          "uid": "99999999-9999-9999-9999-999999999999"}},
        )
        ds.prepare(
            AllocatedClaim(
                ref=ClaimRef(namespace="d", name="c1", uid=UID1),
                devices=[AllocatedDevice(device="gpu-0")],
            )
        )
        mgr = CheckpointCleanupManager(ds, client, interval=3600)
        assert mgr.cleanup_pass() == 1


class _MockVfioTree:
    """Fake /sys/bus/pci + /dev/vfio layout with a functional rebind hook."""

    def __init__(self, root, pci="0000:0c:00.0", group="42"):
        self.root = root
        self.pci = pci
        self.group = group
        self.sysfs = os.path.join(root, "sys")
        self.dev = os.path.join(root, "dev")
        devdir = os.path.join(self.sysfs, "bus", "pci", "devices", pci)
        os.makedirs(devdir)
        for drv in ("amdgpu", "vfio-pci"):
            os.makedirs(os.path.join(self.sysfs, "bus", "pci", "drivers", drv), exist_ok=True)
        os.symlink(
            os.path.join("..", "..", "drivers", "amdgpu"), os.path.join(devdir, "driver")
        )
        gdir = os.path.join(self.sysfs, "kernel", "iommu_groups", group)
        os.makedirs(gdir)
        os.symlink(gdir, os.path.join(devdir, "iommu_group"))
        os.makedirs(os.path.join(self.dev, "vfio"))
        open(os.path.join(self.dev, "vfio", "vfio"), "w").close()
        open(os.path.join(self.dev, "vfio", group), "w").close()

    def rebind(self, pci, driver):
        devdir = os.path.join(self.sysfs, "bus", "pci", "devices", pci)
        os.unlink(os.path.join(devdir, "driver"))
        os.symlink(
            os.path.join("..", "..", "drivers", driver), os.path.join(devdir, "driver")
        )


class TestVfio:
    @pytest.fixture
    def vfio_env(self, tmp_path, env):
        tree, lib = env
        vt = _MockVfioTree(str(tmp_path / "vfio"))
        mgr = VfioPciManager(
            sysfs_root=vt.sysfs,
            dev_root=vt.dev,
            busy_check=lambda pci: False,
            rebind_hook=vt.rebind,
        )
        gpu = lib.gpus()[0]
        gpu.pci_bus_id = vt.pci
        return vt, mgr, gpu

    def test_prepare_rebinds_and_reports_group(self, vfio_env):
        vt, mgr, gpu = vfio_env
        info = mgr.prepare(gpu)
        assert mgr.current_driver(vt.pci) == "vfio-pci"
        assert info.iommu_group == "42"
        assert info.vfio_dev_path.endswith("/vfio/42")
        mgr.unprepare(vt.pci)
        assert mgr.current_driver(vt.pci) == "amdgpu"

    def test_cdi_edits(self, vfio_env):
        vt, mgr, gpu = vfio_env
        info = mgr.prepare(gpu)
        edits = mgr.cdi_edits(info)
        paths = [n.path for n in edits.device_nodes]
        assert "/dev/vfio/vfio" in paths
        assert "/dev/vfio/42" in paths
        assert "AMD_VISIBLE_DEVICES=void" in edits.env

    def test_busy_gpu_times_out(self, vfio_env):
        vt, mgr, gpu = vfio_env
        mgr._busy_check = lambda pci: True
        with pytest.raises(VfioError, match="busy"):
            mgr.wait_for_gpu_free(vt.pci, timeout=0.1)

    def test_sriov_vf_guard(self, vfio_env, tmp_path):
        vt, mgr, gpu = vfio_env
        devdir = os.path.join(vt.sysfs, "bus", "pci", "devices", vt.pci)
        os.symlink(devdir, os.path.join(devdir, "physfn"))
        with pytest.raises(VfioError, match="SR-IOV"):
            mgr.prepare(gpu)

    def _iommufd_cfg(self):
        from k8s_dra_driver_gpu_amd.api.configs import (
            IOMMU_PREFER_IOMMUFD,
            IommuConfig,
            VfioDeviceConfig,
        )

        return VfioDeviceConfig(
            iommu=IommuConfig(backend_policy=IOMMU_PREFER_IOMMUFD, enable_api_device=True)
        )

    def test_iommufd_cdev_discovered_from_sysfs(self, vfio_env):
        """The IOMMUFD cdev index is per-device, NOT the IOMMU group number:
        it must be read from <pci>/vfio-dev/vfioX (ref vfio-device.go)."""
        vt, mgr, gpu = vfio_env
        open(os.path.join(vt.dev, "iommu"), "w").close()
        devdir = os.path.join(vt.sysfs, "bus", "pci", "devices", vt.pci)
        os.makedirs(os.path.join(devdir, "vfio-dev", "vfio7"))
        info = mgr.prepare(gpu, self._iommufd_cfg())
        assert info.vfio_dev_path.endswith("/vfio/devices/vfio7")
        assert "42" not in os.path.basename(info.vfio_dev_path)

    def test_iommufd_missing_cdev_fails_prepare(self, vfio_env):
        """IOMMUFD selected but no vfio-dev entry: prepare must fail loudly
        rather than inject a nonexistent node."""
        vt, mgr, gpu = vfio_env
        open(os.path.join(vt.dev, "iommu"), "w").close()
        with pytest.raises(VfioError, match="vfio cdev"):
            mgr.prepare(gpu, self._iommufd_cfg())


class TestVfioInDeviceState:
    """VFIO wired through the Prepare/Unprepare state machine."""

    @pytest.fixture
    def vfio_state(self, tmp_path, env):
        tree, lib = env
        vt = _MockVfioTree(str(tmp_path / "vfio"), pci=lib.gpus()[0].pci_bus_id)
        mgr = VfioPciManager(
            sysfs_root=vt.sysfs, dev_root=vt.dev,
            busy_check=lambda pci: False, rebind_hook=vt.rebind,
        )
        ds = DeviceState(
            devicelib=lib,
            cdi=CdiHandler(cdi_root=str(tmp_path / "cdi"), dev_root=tree.dev_root),
            checkpoints=CheckpointManager(str(tmp_path / "state")),
            state_dir=str(tmp_path / "state"),
            vfio=mgr,
        )
        return vt, mgr, ds

    def test_vfio_claim_lifecycle(self, vfio_state):
        vt, mgr, ds = vfio_state
        res = ds.prepare(
            AllocatedClaim(
                ref=ClaimRef(namespace="d", name="v", uid=UID1),
                devices=[AllocatedDevice(device="gpu-0-vfio")],
            )
        )
        assert res[0].cdi_device_ids
        assert mgr.current_driver(vt.pci) == "vfio-pci"
        pc = ds.checkpoints.load().get_claim(UID1)
        assert pc.devices[0].type == "vfio"
        assert pc.devices[0].pci_bus_id == vt.pci  # checkpointed for restart
        ds.unprepare(UID1)
        assert mgr.current_driver(vt.pci) == "amdgpu"

    def test_vfio_unprepare_after_restart_uses_checkpointed_pci(self, vfio_state):
        """After a plugin restart the GPU is bound to vfio-pci and has no drm
        card, so UUID lookup fails — unprepare must fall back to the
        checkpointed PCI bus ID instead of silently stranding the device."""
        vt, mgr, ds = vfio_state
        ds.prepare(
            AllocatedClaim(
                ref=ClaimRef(namespace="d", name="v", uid=UID1),
                devices=[AllocatedDevice(device="gpu-0-vfio")],
            )
        )
        assert mgr.current_driver(vt.pci) == "vfio-pci"
        # simulate restart: devicelib can no longer resolve the UUID
        ds.devicelib.gpu_by_uuid = lambda uuid: None
        ds.unprepare(UID1)
        assert mgr.current_driver(vt.pci) == "amdgpu"

    def test_vfio_conflicts_with_whole_gpu(self, vfio_state):
        vt, mgr, ds = vfio_state
        ds.prepare(
            AllocatedClaim(
                ref=ClaimRef(namespace="d", name="v", uid=UID1),
                devices=[AllocatedDevice(device="gpu-0-vfio")],
            )
        )
        from k8s_dra_driver_gpu_amd.plugin.device_state import PrepareError

        uid2 = "22222222-2222-2222-2222-222222222222"
        with pytest.raises(PrepareError, match="VFIO"):
            ds.prepare(
                AllocatedClaim(
                    ref=ClaimRef(namespace="d", name="g", uid=uid2),
                    devices=[AllocatedDevice(device="gpu-0")],
                )
            )

    def test_vfio_gate_off(self, tmp_path, env):
        tree, lib = env
        ds = DeviceState(
            devicelib=lib,
            cdi=CdiHandler(cdi_root=str(tmp_path / "cdi2"), dev_root=tree.dev_root),
            checkpoints=CheckpointManager(str(tmp_path / "state2")),
            state_dir=str(tmp_path / "state2"),
        )
        from k8s_dra_driver_gpu_amd.plugin.device_state import PrepareError

        with pytest.raises(PrepareError, match="PassthroughSupport"):
            ds.prepare(
                AllocatedClaim(
                    ref=ClaimRef(namespace="d", name="v", uid=UID1),
                    devices=[AllocatedDevice(device="gpu-0-vfio")],
                )
            )


class TestOrphanCdiSweep:
    def test_orphaned_spec_removed_live_kept(self, env, tmp_path):
        tree, lib = env
        cdi = CdiHandler(cdi_root=str(tmp_path / "cdi-o"), dev_root=tree.dev_root)
        ds = DeviceState(
            devicelib=lib, cdi=cdi,
            checkpoints=CheckpointManager(str(tmp_path / "state-o")),
            state_dir=str(tmp_path / "state-o"),
        )
        client = FakeClient()
        client.create("resourceclaims",
                      {"metadata": {"name": "c1", "namespace": "d", "uid": UID1}})
        ds.prepare(
            AllocatedClaim(
                ref=ClaimRef(namespace="d", name="c1", uid=UID1),
                devices=[AllocatedDevice(device="gpu-0")],
            )
        )
        # orphan: spec file with no checkpoint entry
        from k8s_dra_driver_gpu_amd.cdi.spec import CdiDevice

        cdi.write_claim_spec("dead0000-0000-0000-0000-000000000000",
                             [CdiDevice(name="zombie")])
        mgr = CheckpointCleanupManager(ds, client, interval=3600)
        removed = mgr.cleanup_pass()
        assert removed == 1
        assert os.path.exists(cdi.claim_spec_path(UID1))  # live claim kept
        assert not os.path.exists(
            cdi.claim_spec_path("dead0000-0000-0000-0000-000000000000")
        )
