"""DRA gRPC contract tests: protobuf wire codec, the plugin service over a
unix socket driven by a fake kubelet client, registration, ResourceSlice
generation, and metrics."""

import os
import threading

import pytest

from k8s_dra_driver_gpu_amd.cdi.spec import CdiHandler
from k8s_dra_driver_gpu_amd.device.devicelib import DeviceLib
from k8s_dra_driver_gpu_amd.device.mock import MockTree
from k8s_dra_driver_gpu_amd.dra import api as dra
from k8s_dra_driver_gpu_amd.dra.protowire import Message, decode_varint, encode_varint
from k8s_dra_driver_gpu_amd.plugin.checkpoint import CheckpointManager, ClaimRef
from k8s_dra_driver_gpu_amd.plugin.device_state import (
    AllocatedClaim,
    AllocatedDevice,
    DeviceState,
)
from k8s_dra_driver_gpu_amd.plugin.driver import GpuDriver, static_claim_resolver
from k8s_dra_driver_gpu_amd.plugin.resourceslice import ResourceSliceGenerator

UID1 = "11111111-1111-1111-1111-111111111111"
UID2 = "22222222-2222-2222-2222-222222222222"


class TestProtowire:
    def test_varint_round_trip(self):
        for v in (0, 1, 127, 128, 300, 2**32, 2**63 - 1):
            buf = encode_varint(v)
            out, pos = decode_varint(buf, 0)
            assert out == v and pos == len(buf)

    def test_message_round_trip(self):
        c = dra.Claim(namespace="ns", uid=UID1, name="claim-a")
        req = dra.NodePrepareResourcesRequest(claims=[c, dra.Claim(name="b")])
        buf = req.to_bytes()
        back = dra.NodePrepareResourcesRequest.from_bytes(buf)
        assert back.claims[0].namespace == "ns"
        assert back.claims[0].uid == UID1
        assert back.claims[1].name == "b"

    def test_map_field_round_trip(self):
        resp = dra.NodePrepareResourcesResponse()
        resp.claims[UID1] = dra.NodePrepareResourceResponse(
            devices=[
                dra.Device(
                    request_names=["r0"],
                    pool_name="node",
                    device_name="gpu-0",
                    cdi_device_ids=["amd.com/gpu=x"],
                )
            ]
        )
        resp.claims[UID2] = dra.NodePrepareResourceResponse(error="boom")
        back = dra.NodePrepareResourcesResponse.from_bytes(resp.to_bytes())
        assert back.claims[UID1].devices[0].device_name == "gpu-0"
        assert back.claims[UID1].devices[0].cdi_device_ids == ["amd.com/gpu=x"]
        assert back.claims[UID2].error == "boom"

    def test_unknown_fields_skipped(self):
        # encode a message with an extra field; decoder must skip it
        from k8s_dra_driver_gpu_amd.dra.protowire import _tag

        extra = _tag(15, 2) + encode_varint(3) + b"xyz"
        buf = dra.Claim(uid=UID1).to_bytes() + extra
        c = dra.Claim.from_bytes(buf)
        assert c.uid == UID1

    def test_empty_message(self):
        assert dra.InfoRequest().to_bytes() == b""
        assert isinstance(dra.InfoRequest.from_bytes(b""), dra.InfoRequest)

    def test_bool_field(self):
        rs = dra.RegistrationStatus(plugin_registered=True, error="")
        back = dra.RegistrationStatus.from_bytes(rs.to_bytes())
        assert back.plugin_registered is True
        rs2 = dra.RegistrationStatus.from_bytes(
            dra.RegistrationStatus(plugin_registered=False, error="e").to_bytes()
        )
        assert rs2.plugin_registered is False and rs2.error == "e"


@pytest.fixture
def served_driver(tmp_path):
    tree = MockTree(root=str(tmp_path / "mock"), num_gpus=2)
    tree.setup()
    lib = DeviceLib(backend=tree.backend())
    state_dir = str(tmp_path / "state")
    ds = DeviceState(
        devicelib=lib,
        cdi=CdiHandler(cdi_root=str(tmp_path / "cdi"), dev_root=tree.dev_root),
        checkpoints=CheckpointManager(state_dir, boot_id="b1"),
        state_dir=state_dir,
    )
    store = {
        UID1: AllocatedClaim(
            ref=ClaimRef(namespace="default", name="c1", uid=UID1),
            devices=[AllocatedDevice(device="gpu-0", request="req0")],
        ),
        UID2: AllocatedClaim(
            ref=ClaimRef(namespace="default", name="c2", uid=UID2),
            devices=[AllocatedDevice(device="gpu-0-cpx-2")],
        ),
    }
    driver = GpuDriver(state=ds, claim_resolver=static_claim_resolver(store), node_name="n1")
    socks = driver.start(plugin_dir=str(tmp_path / "plugin"), registry_dir=str(tmp_path / "reg"))
    yield driver, socks, lib
    driver.stop()


class TestPluginService:
    def test_prepare_unprepare_over_grpc(self, served_driver):
        driver, socks, lib = served_driver
        client = dra.DRAPluginClient(f"unix://{socks['dra']}")
        resp = client.prepare([dra.Claim(namespace="default", name="c1", uid=UID1)])
        r = resp.claims[UID1]
        assert r.error == ""
        assert r.devices[0].device_name == "gpu-0"
        assert r.devices[0].pool_name == "n1"
        assert r.devices[0].cdi_device_ids[0].startswith("amd.com/gpu=")
        uresp = client.unprepare([dra.Claim(namespace="default", name="c1", uid=UID1)])
        assert uresp.claims[UID1].error == ""
        client.close()

    def test_partition_claim_over_grpc(self, served_driver):
        driver, socks, lib = served_driver
        client = dra.DRAPluginClient(f"unix://{socks['dra']}")
        resp = client.prepare([dra.Claim(namespace="default", name="c2", uid=UID2)])
        assert resp.claims[UID2].error == ""
        assert lib.gpu_by_minor(0).compute_partition == "CPX"
        client.unprepare([dra.Claim(namespace="default", name="c2", uid=UID2)])
        assert lib.gpu_by_minor(0).compute_partition == "SPX"
        client.close()

    def test_batch_error_isolation(self, served_driver):
        driver, socks, _ = served_driver
        client = dra.DRAPluginClient(f"unix://{socks['dra']}")
        resp = client.prepare(
            [
                dra.Claim(namespace="default", name="c1", uid=UID1),
                dra.Claim(namespace="default", name="zz", uid="99999999-9999-9999-9999-999999999999"),
            ]
        )
        assert resp.claims[UID1].error == ""
        assert "no allocation" in resp.claims["99999999-9999-9999-9999-999999999999"].error
        client.unprepare([dra.Claim(uid=UID1)])
        client.close()

    def test_v1_package_alias(self, served_driver):
        driver, socks, _ = served_driver
        client = dra.DRAPluginClient(f"unix://{socks['dra']}", package="v1")
        resp = client.prepare([dra.Claim(namespace="default", name="c1", uid=UID1)])
        assert resp.claims[UID1].error == ""
        client.unprepare([dra.Claim(uid=UID1)])
        client.close()

    def test_registration_service(self, served_driver):
        driver, socks, _ = served_driver
        reg = dra.RegistrationClient(f"unix://{socks['registration']}")
        info = reg.get_info()
        assert info.type == "DRAPlugin"
        assert info.name == "gpu.amd.com"
        assert info.endpoint == socks["dra"]
        # The kubelet matches DRA *service identifiers*, not bare versions
        # (ref vendor .../dra/v1beta1/types.go:23).
        assert info.supported_versions == ["v1.DRAPlugin", "v1beta1.DRAPlugin"]
        reg.notify(True)
        assert driver.registration.registered is True
        reg.close()

    def test_metrics_recorded(self, served_driver):
        driver, socks, _ = served_driver
        client = dra.DRAPluginClient(f"unix://{socks['dra']}")
        client.prepare([dra.Claim(namespace="default", name="c1", uid=UID1)])
        client.unprepare([dra.Claim(uid=UID1)])
        client.close()
        from prometheus_client import generate_latest

        text = generate_latest(driver.metrics.registry).decode()
        assert 'amd_dra_requests_total{operation="prepare",status="success"} 1.0' in text
        assert 'amd_dra_requests_total{operation="unprepare",status="success"} 1.0' in text


class TestResourceSlice:
    def _lib(self, tmp_path, n=2):
        tree = MockTree(root=str(tmp_path / "m"), num_gpus=n)
        tree.setup()
        return DeviceLib(backend=tree.backend())

    def test_legacy_slice(self, tmp_path):
        lib = self._lib(tmp_path)
        gen = ResourceSliceGenerator(lib, node_name="n1")
        slices = gen.generate()
        assert len(slices) == 1
        devs = slices[0]["spec"]["devices"]
        assert len(devs) == 2
        d0 = devs[0]["basic"]
        assert d0["attributes"]["productName"]["string"] == "AMD Instinct MI355X"
        assert d0["attributes"]["architecture"]["string"] == "gfx950"
        assert int(d0["capacity"]["memory"]["value"]) == 288 * 1024**3
        assert d0["attributes"]["xgmiHiveID"]["string"].startswith("hive-")

    def test_partitionable_slice_counters(self, tmp_path):
        lib = self._lib(tmp_path, n=1)
        gen = ResourceSliceGenerator(lib, node_name="n1", partitionable=True)
        sl = gen.generate()[0]
        spec = sl["spec"]
        assert len(spec["sharedCounters"]) == 1
        counters = spec["sharedCounters"][0]["counters"]
        assert len([k for k in counters if k.startswith("xcd-")]) == 8
        names = [d["name"] for d in spec["devices"]]
        assert "gpu-0" in names
        # CPX placements: 8; DPX: 2; QPX: 4
        assert len([n for n in names if "-cpx-" in n]) == 8
        assert len([n for n in names if "-dpx-" in n]) == 2
        assert len([n for n in names if "-qpx-" in n]) == 4
        # whole GPU consumes all 8 xcd counters
        gpu0 = next(d for d in spec["devices"] if d["name"] == "gpu-0")
        consumed = gpu0["basic"]["consumesCounters"][0]["counters"]
        assert len([k for k in consumed if k.startswith("xcd-")]) == 8
        # cpx-3 consumes exactly xcd-3
        cpx3 = next(d for d in spec["devices"] if d["name"] == "gpu-0-cpx-3")
        ccons = cpx3["basic"]["consumesCounters"][0]["counters"]
        assert list(k for k in ccons if k.startswith("xcd-")) == ["xcd-3"]

    def test_taints_attached(self, tmp_path):
        lib = self._lib(tmp_path, n=1)
        taint = [{"key": "amd.com/gpu-unhealthy", "effect": "NoSchedule"}]
        gen = ResourceSliceGenerator(lib, node_name="n1", taints={"gpu-0": taint})
        sl = gen.generate()[0]
        gpu0 = next(d for d in sl["spec"]["devices"] if d["name"] == "gpu-0")
        assert gpu0["basic"]["taints"] == taint


class TestDeviceMetadataGate:
    def test_extended_attributes(self, tmp_path):
        tree = MockTree(root=str(tmp_path / "m"), num_gpus=1)
        tree.setup()
        lib = DeviceLib(backend=tree.backend())
        sl = ResourceSliceGenerator(lib, node_name="n1", extended_metadata=True).generate()[0]
        attrs = sl["spec"]["devices"][0]["basic"]["attributes"]
        assert attrs["serial"]["string"].startswith("MOCKSER")
        assert attrs["simdCount"]["int"] == 1024
        assert "vbiosVersion" in attrs
        # off by default
        sl2 = ResourceSliceGenerator(lib, node_name="n1").generate()[0]
        assert "serial" not in sl2["spec"]["devices"][0]["basic"]["attributes"]


class TestPreparedDevicesGauge:
    def test_gauge_tracks_prepared(self, served_driver):
        from prometheus_client import generate_latest

        driver, socks, _ = served_driver
        client = dra.DRAPluginClient(f"unix://{socks['dra']}")
        client.prepare([dra.Claim(namespace="default", name="c1", uid=UID1)])
        text = generate_latest(driver.metrics.registry).decode()
        assert 'amd_dra_prepared_devices{type="gpu"} 1.0' in text
        client.unprepare([dra.Claim(uid=UID1)])
        text = generate_latest(driver.metrics.registry).decode()
        assert 'amd_dra_prepared_devices{type="gpu"} 0.0' in text
        client.close()


class TestVfioAdvertisement:
    def test_vfio_devices_in_slices(self, tmp_path):
        tree = MockTree(root=str(tmp_path / "m"), num_gpus=1)
        tree.setup()
        lib = DeviceLib(backend=tree.backend())
        sl = ResourceSliceGenerator(lib, node_name="n1", vfio=True,
                                    partitionable=True).generate()[0]
        names = [d["name"] for d in sl["spec"]["devices"]]
        assert "gpu-0-vfio" in names
        v = next(d for d in sl["spec"]["devices"] if d["name"] == "gpu-0-vfio")
        assert v["basic"]["attributes"]["type"]["string"] == "vfio"
        # consumes the whole GPU's counters -> scheduler can't co-allocate
        assert v["basic"]["consumesCounters"][0]["counters"]["xcd-0"]
        # gate off by default
        sl2 = ResourceSliceGenerator(lib, node_name="n1").generate()[0]
        assert "gpu-0-vfio" not in [d["name"] for d in sl2["spec"]["devices"]]

    def test_vfio_selectable_by_deviceclass(self, tmp_path):
        import yaml as _yaml

        from k8s_dra_driver_gpu_amd.k8s.celselect import device_matches_class

        tree = MockTree(root=str(tmp_path / "m2"), num_gpus=1)
        tree.setup()
        lib = DeviceLib(backend=tree.backend())
        sl = ResourceSliceGenerator(lib, node_name="n1", vfio=True).generate()[0]
        from k8s_dra_driver_gpu_amd.utils.helmlite import chart_deviceclasses

        chart = os.path.join(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            "deployments", "helm", "amd-dra-driver",
        )
        dcs = {d["metadata"]["name"]: d for d in chart_deviceclasses(chart)}
        v = next(d for d in sl["spec"]["devices"] if d["name"] == "gpu-0-vfio")
        assert device_matches_class(v, "gpu.amd.com", dcs["vfio.gpu.amd.com"])
        assert not device_matches_class(v, "gpu.amd.com", dcs["gpu.amd.com"])


class TestPoolGeneration:
    def test_generation_increments_per_publish(self, tmp_path):
        tree = MockTree(root=str(tmp_path / "m"), num_gpus=1)
        tree.setup()
        lib = DeviceLib(backend=tree.backend())
        gen = ResourceSliceGenerator(lib, node_name="n1")
        g1 = gen.generate()[0]["spec"]["pool"]["generation"]
        g2 = gen.generate()[0]["spec"]["pool"]["generation"]
        assert g2 == g1 + 1
