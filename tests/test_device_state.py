"""DeviceState (Prepare/Unprepare) tests against the mock device layer.

Mirrors the reference's device_state_test.go scenarios: prepare idempotency
and checkpoint round-trip (:374-430), unprepare-missing-claim no-op,
overlapping-device guard (:230), config precedence (:73,142), partial-prepare
rollback, startup reconciliation of unknown partitions."""

import json
import os

import pytest

from k8s_dra_driver_gpu_amd.api.configs import APIVERSION, CPX, SPX
from k8s_dra_driver_gpu_amd.cdi.spec import CdiHandler
from k8s_dra_driver_gpu_amd.device.devicelib import DeviceLib, PartitionSpec
from k8s_dra_driver_gpu_amd.device.mock import MockTree
from k8s_dra_driver_gpu_amd.plugin.checkpoint import (
    PREPARE_COMPLETED,
    PREPARE_STARTED,
    CheckpointManager,
    ClaimRef,
    PreparedClaim,
    PreparedDevice,
)
from k8s_dra_driver_gpu_amd.plugin.device_state import (
    AllocatedClaim,
    AllocatedDevice,
    DeviceState,
    PrepareError,
)

UID1 = "11111111-1111-1111-1111-111111111111"
UID2 = "22222222-2222-2222-2222-222222222222"


@pytest.fixture
def env(tmp_path):
    tree = MockTree(root=str(tmp_path / "mock"), num_gpus=2)
    tree.setup()
    lib = DeviceLib(backend=tree.backend())
    state_dir = str(tmp_path / "state")
    cdi = CdiHandler(cdi_root=str(tmp_path / "cdi"), dev_root=tree.dev_root)
    cps = CheckpointManager(state_dir, boot_id="boot-1")
    ds = DeviceState(devicelib=lib, cdi=cdi, checkpoints=cps, state_dir=state_dir)
    return tree, lib, cdi, cps, ds


def claim(uid, *devices, configs=None):
    return AllocatedClaim(
        ref=ClaimRef(namespace="default", name=f"claim-{uid[:4]}", uid=uid),
        devices=[AllocatedDevice(device=d, configs=configs or []) for d in devices],
    )


class TestPrepareGpu:
    def test_whole_gpu(self, env):
        tree, lib, cdi, cps, ds = env
        results = ds.prepare(claim(UID1, "gpu-0"))
        assert len(results) == 1
        assert results[0].cdi_device_ids == [f"amd.com/gpu=claim-{UID1}-gpu-0"]
        # CDI spec exists and injects kfd + render node
        spec = json.load(open(cdi.claim_spec_path(UID1)))
        nodes = spec["devices"][0]["containerEdits"]["deviceNodes"]
        paths = [n["path"] for n in nodes]
        assert "/dev/kfd" in paths
        assert "/dev/dri/renderD128" in paths
        assert "/dev/dri/card0" in paths
        # checkpointed as completed
        pc = cps.load().get_claim(UID1)
        assert pc.state == PREPARE_COMPLETED
        assert pc.devices[0].name == "gpu-0"

    def test_idempotent(self, env):
        _, _, _, _, ds = env
        r1 = ds.prepare(claim(UID1, "gpu-0"))
        r2 = ds.prepare(claim(UID1, "gpu-0"))
        assert [r.cdi_device_ids for r in r1] == [r.cdi_device_ids for r in r2]

    def test_unknown_gpu(self, env):
        _, _, _, _, ds = env
        with pytest.raises(PrepareError, match="no GPU"):
            ds.prepare(claim(UID1, "gpu-9"))

    def test_two_claims_share_gpu_timeslicing(self, env):
        """gpu-test2 analog: sharing needs no device mutation; two claims on
        one GPU both succeed (TimeSlicing default)."""
        _, _, _, _, ds = env
        ds.prepare(claim(UID1, "gpu-0"))
        with pytest.raises(PrepareError, match="already prepared"):
            ds.prepare(claim(UID2, "gpu-0"))
        # NOTE: DRA-level sharing is one claim consumed by multiple pods /
        # containers; a second *claim* on the same device is double-allocation.

    def test_sharing_env(self, env):
        _, _, cdi, _, ds = env
        cfg = {
            "apiVersion": APIVERSION,
            "kind": "GpuConfig",
            "sharing": {"strategy": "TimeSlicing", "timeSlicingConfig": {"interval": "Long"}},
        }
        ds.prepare(claim(UID1, "gpu-0", configs=[cfg]))
        spec = json.load(open(cdi.claim_spec_path(UID1)))
        env_list = spec["devices"][0]["containerEdits"]["env"]
        assert "AMDDRA_SHARING=TimeSlicing:Long" in env_list
        # honest sharing: no pseudo-knob env vars — every non-AMDDRA var we
        # emit must be documented ROCm surface
        for e in env_list:
            assert e.split("=")[0].startswith(("AMDDRA_", "ROC_")), e

    def _spatial_cfg(self, **spc):
        return {
            "apiVersion": APIVERSION,
            "kind": "GpuConfig",
            "sharing": {
                "strategy": "SpatialPartitioning",
                "spatialPartitioningConfig": spc,
            },
        }

    def test_spatial_sharing_emits_documented_cu_mask(self, env):
        """xcdCount=2 -> ROC_GLOBAL_CU_MASK (documented HIP env var) with the
        low 64 bits set (2 XCDs x 32 CUs on MI355X)."""
        _, _, cdi, _, ds = env
        ds.prepare(claim(UID1, "gpu-0", configs=[self._spatial_cfg(xcdCount=2)]))
        spec = json.load(open(cdi.claim_spec_path(UID1)))
        env_list = spec["devices"][0]["containerEdits"]["env"]
        assert "AMDDRA_SHARING=Spatial:xcd=2" in env_list
        assert f"ROC_GLOBAL_CU_MASK={hex((1 << 64) - 1)}" in env_list

    def test_spatial_full_chip_emits_no_mask(self, env):
        _, _, cdi, _, ds = env
        ds.prepare(claim(UID1, "gpu-0", configs=[self._spatial_cfg(xcdCount=8)]))
        spec = json.load(open(cdi.claim_spec_path(UID1)))
        env_list = spec["devices"][0]["containerEdits"]["env"]
        assert not any(e.startswith("ROC_GLOBAL_CU_MASK") for e in env_list)

    def test_spatial_percentage_rounds_to_xcd_granularity(self, env):
        _, _, cdi, _, ds = env
        ds.prepare(
            claim(UID1, "gpu-0", configs=[self._spatial_cfg(defaultXcdPercentage=50)])
        )
        spec = json.load(open(cdi.claim_spec_path(UID1)))
        env_list = spec["devices"][0]["containerEdits"]["env"]
        assert "AMDDRA_SHARING=Spatial:xcd=4" in env_list
        assert f"ROC_GLOBAL_CU_MASK={hex((1 << 128) - 1)}" in env_list


class TestPreparePartition:
    def test_partition_prepare(self, env):
        tree, lib, cdi, cps, ds = env
        r = ds.prepare(claim(UID1, "gpu-0-cpx-3"))
        assert len(r) == 1
        g = lib.gpu_by_minor(0)
        assert g.compute_partition == CPX
        pc = cps.load().get_claim(UID1)
        assert pc.devices[0].type == "partition"
        assert pc.devices[0].partition_index == 3

    def test_partition_unprepare_resets_mode(self, env):
        tree, lib, _, _, ds = env
        ds.prepare(claim(UID1, "gpu-0-cpx-3"))
        ds.unprepare(UID1)
        assert lib.gpu_by_minor(0).compute_partition == SPX

    def test_partition_mode_kept_while_other_claims(self, env):
        tree, lib, _, _, ds = env
        ds.prepare(claim(UID1, "gpu-0-cpx-0"))
        ds.prepare(claim(UID2, "gpu-0-cpx-1"))
        ds.unprepare(UID1)
        assert lib.gpu_by_minor(0).compute_partition == CPX  # UID2 still holds one
        ds.unprepare(UID2)
        assert lib.gpu_by_minor(0).compute_partition == SPX

    def test_eight_concurrent_partitions(self, env):
        """BASELINE.json config 4: 8 single-partition claims on 1 GPU."""
        tree, lib, _, _, ds = env
        uids = [f"{i}{i}{i}{i}{i}{i}{i}{i}-0000-0000-0000-00000000000{i}" for i in range(8)]
        for i, uid in enumerate(uids):
            ds.prepare(claim(uid, f"gpu-0-cpx-{i}"))
        assert lib.gpu_by_minor(0).compute_partition == CPX
        for uid in uids:
            ds.unprepare(uid)
        assert lib.gpu_by_minor(0).compute_partition == SPX


class TestOverlapGuard:
    def test_gpu_then_partition(self, env):
        _, _, _, _, ds = env
        ds.prepare(claim(UID1, "gpu-0"))
        with pytest.raises(PrepareError, match="prepared whole"):
            ds.prepare(claim(UID2, "gpu-0-cpx-0"))

    def test_partition_then_gpu(self, env):
        _, _, _, _, ds = env
        ds.prepare(claim(UID1, "gpu-0-cpx-0"))
        with pytest.raises(PrepareError, match="has partitions prepared"):
            ds.prepare(claim(UID2, "gpu-0"))

    def test_same_partition_twice(self, env):
        _, _, _, _, ds = env
        ds.prepare(claim(UID1, "gpu-0-cpx-0"))
        with pytest.raises(PrepareError, match="already prepared"):
            ds.prepare(claim(UID2, "gpu-0-cpx-0"))

    def test_different_gpu_ok(self, env):
        _, _, _, _, ds = env
        ds.prepare(claim(UID1, "gpu-0"))
        ds.prepare(claim(UID2, "gpu-8"))  # no error (second GPU is card minor 8)


class TestUnprepare:
    def test_missing_claim_noop(self, env):
        _, _, _, _, ds = env
        ds.unprepare(UID1)  # no error

    def test_removes_spec_and_checkpoint(self, env):
        _, _, cdi, cps, ds = env
        ds.prepare(claim(UID1, "gpu-0"))
        assert os.path.exists(cdi.claim_spec_path(UID1))
        ds.unprepare(UID1)
        assert not os.path.exists(cdi.claim_spec_path(UID1))
        assert cps.load().get_claim(UID1) is None


class TestCrashRecovery:
    def test_partial_prepare_rollback(self, env):
        """A claim stuck in PrepareStarted is rolled back and re-prepared."""
        tree, lib, cdi, cps, ds = env
        # simulate crash: PrepareStarted entry with a partition device recorded
        cps.update(
            lambda d: d.set_claim(
                UID1,
                PreparedClaim(
                    state=PREPARE_STARTED,
                    claim=ClaimRef(namespace="default", name="c", uid=UID1),
                    devices=[
                        PreparedDevice(
                            type="partition",
                            name="gpu-0-cpx-0",
                            parent_uuid=lib.gpus()[0].uuid,
                            compute_mode=CPX,
                            partition_index=0,
                        )
                    ],
                ),
            )
        )
        lib.backend.set_compute_partition(0, CPX)
        lib.invalidate()
        r = ds.prepare(claim(UID1, "gpu-0"))  # now claim it whole
        assert len(r) == 1
        assert lib.gpu_by_minor(0).compute_partition == SPX

    def test_randomized_crash_injection_recovers(self, env):
        """Seeded fuzz: fail the CDI claim-spec write or the partition write
        at a random prepare, verify the claim fails atomically (no
        checkpoint residue, device back in SPX), then a clean retry
        succeeds and unprepare leaves nothing behind."""
        import random

        tree, lib, cdi, cps, ds = env
        rng = random.Random(777)
        for trial in range(10):
            uid = f"{trial:08d}-dsfz-4000-8000-000000000000"
            dev = rng.choice(["gpu-0", "gpu-0-cpx-3", "gpu-8"])
            target = rng.choice(["cdi", "sysfs"])
            real_write = cdi.write_claim_spec
            real_set = lib.backend.set_compute_partition

            def boom(*a, **k):
                raise OSError("injected crash")

            if target == "cdi":
                cdi.write_claim_spec = boom
            elif "cpx" in dev:
                lib.backend.set_compute_partition = boom
            try:
                if target == "cdi" or "cpx" in dev:
                    with pytest.raises(Exception):
                        ds.prepare(claim(uid, dev))
                else:
                    ds.prepare(claim(uid, dev))
                    ds.unprepare(uid)
                    continue
            finally:
                cdi.write_claim_spec = real_write
                lib.backend.set_compute_partition = real_set
            # atomic failure: no completed checkpoint entry survives
            pc = cps.load().get_claim(uid)
            assert pc is None or pc.state == PREPARE_STARTED
            lib.invalidate()
            # clean retry
            r = ds.prepare(claim(uid, dev))
            assert len(r) == 1
            ds.unprepare(uid)
            assert cps.load().get_claim(uid) is None
            assert not os.path.exists(cdi.claim_spec_path(uid))
            lib.invalidate()
            assert lib.gpu_by_minor(0).compute_partition == SPX

    def test_destroy_unknown_partitions(self, env):
        tree, lib, _, _, ds = env
        lib.backend.set_compute_partition(8, CPX)  # second GPU = card minor 8
        lib.invalidate()
        assert ds.destroy_unknown_partitions() == 1
        assert lib.gpu_by_minor(8).compute_partition == SPX

    def test_known_partitions_kept(self, env):
        tree, lib, _, _, ds = env
        ds.prepare(claim(UID1, "gpu-0-cpx-0"))
        assert ds.destroy_unknown_partitions() == 0
        assert lib.gpu_by_minor(0).compute_partition == CPX


class TestPartitionSharing:
    def test_partition_with_timeslicing(self, env):
        """Partition claim + TimeSlicing sharing combined (ref DynMIG
        '1 MIG + TimeSlicing config' scenario): the partition is created AND
        the sharing env lands in its CDI spec."""
        tree, lib, cdi, _, ds = env
        cfg = {
            "apiVersion": APIVERSION,
            "kind": "PartitionConfig",
            "sharing": {"strategy": "TimeSlicing",
                        "timeSlicingConfig": {"interval": "Short"}},
        }
        ds.prepare(claim(UID1, "gpu-0-cpx-2", configs=[cfg]))
        assert lib.gpu_by_minor(0).compute_partition == CPX
        spec = json.load(open(cdi.claim_spec_path(UID1)))
        env_list = spec["devices"][0]["containerEdits"]["env"]
        assert "AMDDRA_SHARING=TimeSlicing:Short" in env_list

    def test_spatial_sharing_on_partition_rejected(self, env):
        _, _, _, cps, ds = env
        cfg = {
            "apiVersion": APIVERSION,
            "kind": "PartitionConfig",
            "sharing": {"strategy": "SpatialPartitioning"},
        }
        with pytest.raises(PrepareError, match="already-partitioned"):
            ds.prepare(claim(UID1, "gpu-0-cpx-2", configs=[cfg]))
        assert cps.load().get_claim(UID1) is None


class TestConfigPrecedence:
    def test_partition_config_on_gpu_rejected(self, env):
        _, _, _, _, ds = env
        cfg = {"apiVersion": APIVERSION, "kind": "PartitionConfig"}
        with pytest.raises(PrepareError, match="whole-GPU"):
            ds.prepare(claim(UID1, "gpu-0", configs=[cfg]))

    def test_gpu_config_on_partition_rejected(self, env):
        _, _, _, _, ds = env
        cfg = {"apiVersion": APIVERSION, "kind": "GpuConfig"}
        with pytest.raises(PrepareError, match="partition"):
            ds.prepare(claim(UID1, "gpu-0-cpx-0", configs=[cfg]))

    def test_invalid_config_rejected_before_mutation(self, env):
        tree, lib, _, cps, ds = env
        cfg = {
            "apiVersion": APIVERSION,
            "kind": "GpuConfig",
            "sharing": {"strategy": "Bogus"},
        }
        with pytest.raises(PrepareError):
            ds.prepare(claim(UID1, "gpu-0", configs=[cfg]))
        assert cps.load().get_claim(UID1) is None  # phase-1 rolled back


class TestMemoryModeClaims:
    def test_nps2_via_partition_config(self, env):
        tree, lib, _, cps, ds = env
        cfg = {"apiVersion": APIVERSION, "kind": "PartitionConfig", "memoryMode": "NPS2"}
        ds.prepare(claim(UID1, "gpu-0-cpx-0", configs=[cfg]))
        g = lib.gpu_by_minor(0)
        assert g.compute_partition == CPX
        assert g.memory_partition == "NPS2"
        pc = cps.load().get_claim(UID1)
        assert pc.devices[0].memory_mode == "NPS2"
        ds.unprepare(UID1)
        g = lib.gpu_by_minor(0)
        assert g.compute_partition == SPX
        assert g.memory_partition == "NPS1"

    def test_invalid_memory_mode_rejected(self, env):
        _, _, _, _, ds = env
        cfg = {"apiVersion": APIVERSION, "kind": "PartitionConfig", "memoryMode": "NPS9"}
        with pytest.raises(PrepareError, match="memoryMode"):
            ds.prepare(claim(UID1, "gpu-0-cpx-0", configs=[cfg]))

    def test_nps2_needs_multiway_split(self, env):
        """memoryMode on a whole-GPU-equivalent split is rejected by the
        device layer compatibility matrix."""
        _, _, _, _, ds = env
        cfg = {"apiVersion": APIVERSION, "kind": "PartitionConfig", "memoryMode": "NPS4"}
        # mock advertises NPS1,NPS2 only -> NPS4 is rejected by the backend
        with pytest.raises(PrepareError):
            ds.prepare(claim(UID1, "gpu-0-dpx-0", configs=[cfg]))


class TestSharedStateDir:
    def test_two_plugin_instances_share_node_state(self, env, tmp_path):
        """Two DeviceState instances over ONE state dir (two driver pods on a
        node, ref driver.go:373-418 node-global pu.lock): churn concurrently,
        overlap guard holds across instances via the shared checkpoint."""
        import threading
        import uuid as uuidlib

        tree, lib, cdi, cps, ds1 = env
        from k8s_dra_driver_gpu_amd.plugin.checkpoint import CheckpointManager

        ds2 = DeviceState(
            devicelib=lib, cdi=cdi,
            checkpoints=CheckpointManager(cps.state_dir, boot_id="boot-1"),
            state_dir=cps.state_dir,
        )
        errors, wins = [], []

        def churn(ds, n):
            for _ in range(n):
                uid = str(uuidlib.uuid4())
                try:
                    ds.prepare(claim(uid, "gpu-0"))
                    wins.append(uid)
                    ds.unprepare(uid)
                except PrepareError:
                    pass  # lost the race for the device: correct
                except Exception as e:  # noqa: BLE001
                    errors.append(e)

        t1 = threading.Thread(target=churn, args=(ds1, 10))
        t2 = threading.Thread(target=churn, args=(ds2, 10))
        t1.start(); t2.start(); t1.join(60); t2.join(60)
        assert not errors, errors[:3]
        assert wins
        assert ds1.prepared_claims() == {}
        assert ds2.prepared_claims() == {}


class TestPrepareLockTimeout:
    def test_contended_pu_lock_times_out(self, env, tmp_path):
        """Another driver pod holding the node pu.lock: prepare fails with a
        timeout within ~prepare_timeout instead of hanging (ref driver.go:381
        10 s lock acquisition)."""
        import subprocess
        import sys
        import time as _t

        tree, lib, cdi, cps, ds = env
        ds.prepare_timeout = 0.5
        lock_path = ds._pu_lock.path
        holder = subprocess.Popen(
            [sys.executable, "-c", f"""
import fcntl, time
fd = open({lock_path!r}, 'w')
fcntl.flock(fd, fcntl.LOCK_EX)
time.sleep(30)
"""],
        )
        try:
            _t.sleep(0.5)  # let the holder grab it
            t0 = _t.monotonic()
            with pytest.raises(Exception, match="lock|timed out"):
                ds.prepare(claim(UID1, "gpu-0"))
            assert _t.monotonic() - t0 < 5.0
        finally:
            holder.kill()
            holder.wait(timeout=5)
        # after the holder dies, prepare succeeds (crash-safe flock)
        ds.prepare_timeout = 10.0
        assert ds.prepare(claim(UID1, "gpu-0"))
