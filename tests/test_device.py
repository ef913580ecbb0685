"""Device-layer tests against the mock MI355X sysfs tree.

Mirrors the reference's nvlib-level coverage (enumeration, MIG
create/find/delete analogs) using the CPU-only mock harness (§4.4 of
SURVEY.md).
"""

import os

import pytest

from k8s_dra_driver_gpu_amd.api.configs import CPX, SPX
from k8s_dra_driver_gpu_amd.device.devicelib import DeviceError, DeviceLib, PartitionSpec
from k8s_dra_driver_gpu_amd.device.info import (
    MI355X_VRAM_BYTES,
    format_partition_name,
    parse_partition_name,
)
from k8s_dra_driver_gpu_amd.device.mock import MockTree


@pytest.fixture
def tree(tmp_path):
    t = MockTree(root=str(tmp_path), num_gpus=8)
    t.setup()
    return t


@pytest.fixture
def lib(tree):
    return DeviceLib(backend=tree.backend())


class TestEnumeration:
    def test_eight_gpus(self, lib):
        gpus = lib.gpus()
        assert len(gpus) == 8
        g0 = gpus[0]
        assert g0.product_name == "AMD Instinct MI355X"
        assert g0.architecture == "gfx950"
        assert g0.gfx_target_version == "9.5.0"
        assert g0.vram_bytes == MI355X_VRAM_BYTES
        assert g0.compute_partition == "SPX"
        assert g0.memory_partition == "NPS1"
        assert g0.pci_bus_id == "0000:0c:00.0"
        assert g0.uuid != "" and g0.uuid != gpus[1].uuid
        assert g0.render_path.endswith("/dri/renderD128")
        assert g0.driver_version == "6.14.5"
        assert g0.simd_count == 1024

    def test_xgmi_topology(self, lib):
        topo = lib.topology()
        gpus = lib.gpus()
        # full mesh: 7 peers each
        for g in gpus:
            peers = topo.links.get(g.uuid, [])
            assert len(peers) == 7, f"{g.canonical_name} has {len(peers)} xGMI peers"
            assert g.xgmi_link_count == 7
        # one hive, one clique id
        cids = {topo.clique_id_for(g.uuid) for g in gpus}
        assert len(cids) == 1
        assert list(cids)[0].startswith("hive-")

    def test_single_gpu_no_hive(self, tmp_path):
        t = MockTree(root=str(tmp_path), num_gpus=1)
        t.setup()
        lib = DeviceLib(backend=t.backend())
        gpus = lib.gpus()
        assert len(gpus) == 1
        assert lib.topology().clique_id_for(gpus[0].uuid) == ""

    def test_dev_nodes_exist(self, tree, lib):
        b = lib.backend
        assert os.path.exists(b.kfd_dev_path())
        for g in lib.gpus():
            assert os.path.exists(g.render_path.replace("/dev/", tree.dev_root + "/", 1)) or \
                os.path.exists(os.path.join(tree.dev_root, "dri", f"renderD{g.render_minor}"))


class TestPartitionNameCodec:
    def test_round_trip(self):
        name = format_partition_name(3, "CPX", 5)
        assert name == "gpu-3-cpx-5"
        assert parse_partition_name(name) == (3, "CPX", 5)

    def test_parse_invalid(self):
        assert parse_partition_name("gpu-3") is None
        assert parse_partition_name("mig-3-cpx-1") is None
        assert parse_partition_name("gpu-x-cpx-1") is None


class TestPartitionLifecycle:
    def test_possible_partitions(self, lib):
        g = lib.gpus()[0]
        specs = lib.possible_partitions(g)
        cpx = [s for s in specs if s.compute_mode == CPX]
        assert len(cpx) == 8  # 8 XCDs -> 8 CPX partitions

    def test_create_cpx_partition(self, lib):
        g = lib.gpus()[0]
        part = lib.create_partition(PartitionSpec(g.uuid, CPX, 3))
        assert part.index == 3
        assert part.compute_mode == CPX
        assert part.vram_bytes == MI355X_VRAM_BYTES // 8
        # parent is now in CPX; all 8 partitions findable
        g2 = lib.gpu_by_uuid(g.uuid)
        assert g2.compute_partition == CPX
        for i in range(8):
            p = lib.find_partition(PartitionSpec(g.uuid, CPX, i))
            assert p is not None, f"partition {i} missing"
            assert p.render_minor >= 0
        # other GPUs untouched
        assert lib.gpus()[1].compute_partition == SPX

    def test_create_idempotent_same_mode(self, lib):
        g = lib.gpus()[0]
        p1 = lib.create_partition(PartitionSpec(g.uuid, CPX, 0))
        p2 = lib.create_partition(PartitionSpec(g.uuid, CPX, 1))
        assert p1.parent_uuid == p2.parent_uuid
        assert p1.render_minor != p2.render_minor

    def test_invalid_index(self, lib):
        g = lib.gpus()[0]
        with pytest.raises(DeviceError, match="out of range"):
            lib.create_partition(PartitionSpec(g.uuid, CPX, 8))

    def test_unsupported_mode(self, lib):
        g = lib.gpus()[0]
        with pytest.raises(DeviceError, match="does not support|unknown"):
            lib.create_partition(PartitionSpec(g.uuid, "TPX", 0))

    def test_reset_partition_mode(self, lib):
        g = lib.gpus()[0]
        lib.create_partition(PartitionSpec(g.uuid, CPX, 0))
        assert lib.gpu_by_uuid(g.uuid).compute_partition == CPX
        assert lib.maybe_reset_partition_mode(g.uuid) is True
        assert lib.gpu_by_uuid(g.uuid).compute_partition == SPX
        assert lib.maybe_reset_partition_mode(g.uuid) is False

    def test_nps2_memory_mode(self, lib):
        g = lib.gpus()[0]
        part = lib.create_partition(PartitionSpec(g.uuid, CPX, 0), memory_mode="NPS2")
        assert part.memory_mode == "NPS2"
        g2 = lib.gpu_by_uuid(g.uuid)
        assert g2.memory_partition == "NPS2"
        lib.maybe_reset_partition_mode(g.uuid)
        assert lib.gpu_by_uuid(g.uuid).memory_partition == "NPS1"

    def test_unknown_gpu(self, lib):
        with pytest.raises(DeviceError, match="no GPU"):
            lib.create_partition(PartitionSpec("nope", CPX, 0))

    def test_cache_invalidation(self, lib):
        g = lib.gpus()[0]
        # cached view stays until invalidation
        lib.create_partition(PartitionSpec(g.uuid, CPX, 0))
        parts = lib.live_partitions()
        assert len([p for p in parts if p.parent_uuid == g.uuid]) == 7  # index 1..7 extra cards


class TestQuiesceCheck:
    """Busy-GPU rejection of partition switches (the reference's in-use
    branch, nvlib.go:1472-1506) via the KFD proc VRAM map."""

    def test_busy_gpu_rejects_switch(self, tree, lib):
        from k8s_dra_driver_gpu_amd.device.devicelib import DeviceBusyError

        g = lib.gpus()[0]
        tree.add_kfd_process(4242, 0)
        with pytest.raises(DeviceBusyError, match="busy .*4242"):
            lib.create_partition(PartitionSpec(g.uuid, CPX, 0))
        # quiesce -> switch succeeds
        tree.remove_kfd_process(4242)
        part = lib.create_partition(PartitionSpec(g.uuid, CPX, 0))
        assert part.compute_mode == CPX

    def test_busy_gpu_rejects_reset(self, tree, lib):
        from k8s_dra_driver_gpu_amd.device.devicelib import DeviceBusyError

        g = lib.gpus()[0]
        lib.create_partition(PartitionSpec(g.uuid, CPX, 0))
        tree.add_kfd_process(777, 0)
        with pytest.raises(DeviceBusyError, match="busy"):
            lib.maybe_reset_partition_mode(g.uuid)
        tree.remove_kfd_process(777)
        assert lib.maybe_reset_partition_mode(g.uuid) is True

    def test_other_gpu_process_does_not_block(self, tree, lib):
        g0, g1 = lib.gpus()[0], lib.gpus()[1]
        tree.add_kfd_process(555, 1)  # busy on a DIFFERENT GPU
        part = lib.create_partition(PartitionSpec(g0.uuid, CPX, 0))
        assert part.compute_mode == CPX

    def test_busy_pids_listing(self, tree, lib):
        tree.add_kfd_process(100, 0)
        tree.add_kfd_process(200, 0)
        tree.add_kfd_process(300, 1)
        b = lib.backend
        g = lib.gpus()[0]
        assert b.gpu_busy_pids(g.minor) == [100, 200]

    def test_unreadable_proc_tree_is_not_busy(self, tree, lib):
        # no proc dir at all -> [] (kernel EBUSY remains the backstop)
        g = lib.gpus()[0]
        assert lib.backend.gpu_busy_pids(g.minor) == []


class TestPartitionWriteFallbacks:
    """Fault-injection matrix for the three-stage partition write path
    (VERDICT round-1 item 4): sysfs EROFS -> amdsmi lib -> amd-smi CLI."""

    def _backend(self, tree):
        # plain SysfsBackend over the mock tree (NOT MockSysfsBackend), so
        # the real fallback chain runs
        from k8s_dra_driver_gpu_amd.device.sysfs import SysfsBackend

        return SysfsBackend(sysfs_root=tree.sysfs_root, dev_root=tree.dev_root)

    def test_sysfs_write_success_short_circuits(self, tree):
        b = self._backend(tree)
        called = {"amdsmi": 0}
        b._amdsmi_set_compute_partition = lambda m, mode: called.__setitem__("amdsmi", 1)
        b.set_compute_partition(0, "CPX")
        assert called["amdsmi"] == 0
        assert b.get_compute_partition(0) == "CPX"

    def test_erofs_falls_back_to_amdsmi(self, tree, monkeypatch):
        import errno

        b = self._backend(tree)
        monkeypatch.setattr(
            b, "_write_sysfs_partition",
            lambda p, m: (_ for _ in ()).throw(OSError(errno.EROFS, "read-only fs")),
        )
        calls = []
        monkeypatch.setattr(
            b, "_amdsmi_set_compute_partition", lambda minor, mode: calls.append(mode) or True
        )
        b.set_compute_partition(0, "CPX")
        assert calls == ["CPX"]

    def test_amdsmi_failure_falls_back_to_cli(self, tree, monkeypatch):
        import errno

        b = self._backend(tree)
        monkeypatch.setattr(
            b, "_write_sysfs_partition",
            lambda p, m: (_ for _ in ()).throw(OSError(errno.EROFS, "read-only fs")),
        )
        monkeypatch.setattr(b, "_amdsmi_set_compute_partition", lambda minor, mode: False)
        calls = []
        monkeypatch.setattr(
            b, "_amdsmi_cli_set_compute_partition", lambda minor, mode: calls.append(mode) or True
        )
        b.set_compute_partition(0, "CPX")
        assert calls == ["CPX"]

    def test_all_arms_fail_raises(self, tree, monkeypatch):
        import errno

        from k8s_dra_driver_gpu_amd.device.sysfs import SysfsError

        b = self._backend(tree)
        monkeypatch.setattr(
            b, "_write_sysfs_partition",
            lambda p, m: (_ for _ in ()).throw(OSError(errno.EROFS, "read-only fs")),
        )
        monkeypatch.setattr(b, "_amdsmi_set_compute_partition", lambda minor, mode: False)
        monkeypatch.setattr(b, "_amdsmi_cli_set_compute_partition", lambda minor, mode: False)
        with pytest.raises(SysfsError, match="failed on card0"):
            b.set_compute_partition(0, "CPX")

    def test_ebusy_is_terminal_no_fallback(self, tree, monkeypatch):
        """EBUSY means 'quiesce first': amdsmi/CLI would hit the same wall,
        so the error surfaces immediately with the quiesce hint."""
        import errno

        from k8s_dra_driver_gpu_amd.device.sysfs import SysfsError

        b = self._backend(tree)
        monkeypatch.setattr(
            b, "_write_sysfs_partition",
            lambda p, m: (_ for _ in ()).throw(OSError(errno.EBUSY, "device busy")),
        )
        calls = []
        monkeypatch.setattr(
            b, "_amdsmi_set_compute_partition", lambda minor, mode: calls.append(1) or True
        )
        with pytest.raises(SysfsError, match="busy"):
            b.set_compute_partition(0, "CPX")
        assert calls == []


class TestAcceleratorProfiles:
    """amd-smi accelerator-partition profile parsing (the MIG
    profile/placement inspection analog, nvlib.go:1202-1277) against REAL
    CLI output captured on an MI355X (tests/fixtures/)."""

    def _fixture(self):
        p = os.path.join(os.path.dirname(__file__), "fixtures",
                         "amd_smi_partition_accelerator.txt")
        return open(p).read()

    def test_parse_real_capture(self):
        from k8s_dra_driver_gpu_amd.device.acceleratorprofiles import (
            parse_accelerator_profiles,
        )

        profs = parse_accelerator_profiles(self._fixture())
        assert 0 in profs
        by_type = {p.type: p for p in profs[0]}
        assert by_type["SPX"].current is True
        assert by_type["SPX"].num_partitions == 1
        assert by_type["SPX"].resources["XCC"] == (8, 1)
        assert by_type["SPX"].resources["JPEG"] == (40, 1)
        assert by_type["DPX"].num_partitions == 2
        assert by_type["DPX"].current is False
        assert by_type["QPX"].num_partitions == 4
        assert by_type["CPX"].num_partitions == 8
        assert by_type["SPX"].memory_caps == ["NPS1"]

    def test_profile_for_mode(self):
        from k8s_dra_driver_gpu_amd.device.acceleratorprofiles import (
            parse_accelerator_profiles,
            profile_for_mode,
        )

        profs = parse_accelerator_profiles(self._fixture())[0]
        assert profile_for_mode(profs, "dpx").type == "DPX"
        assert profile_for_mode(profs, "cpx").num_partitions == 8
        assert profile_for_mode(profs, "TPX") is None

    def test_missing_cli_returns_none(self):
        from k8s_dra_driver_gpu_amd.device.acceleratorprofiles import (
            read_accelerator_profiles,
        )

        assert read_accelerator_profiles(cli="/nonexistent/amd-smi") is None

    def test_truncated_output_tolerated(self):
        from k8s_dra_driver_gpu_amd.device.acceleratorprofiles import (
            parse_accelerator_profiles,
        )

        text = self._fixture()
        assert parse_accelerator_profiles(text[: len(text) // 2])
        assert parse_accelerator_profiles("") == {}
        assert parse_accelerator_profiles("garbage\nlines\n") == {}
