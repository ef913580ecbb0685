#!/bin/bash
# Production soak: full stack live on a real MI355X for ~SOAK_S seconds —
# continuous claim churn, fabricd-backed ComputeDomain held Ready, periodic
# HBM/MFMA probes and health polls. Writes a per-minute log + summary.
set -u
cd /root/repo
export TMPDIR=/tmp
OUT=gpurun_out/soak
mkdir -p "$OUT"
SOAK_S=${SOAK_S:-600}
# write directly to the file: a piped tee can hang on EOF if a
# supervised fabricd child inherits the pipe
timeout $((SOAK_S + 180)) python -u - > "$OUT/soak.txt" 2>&1 <<PYEOF
import json, os, statistics, tempfile, time, uuid
from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster
from k8s_dra_driver_gpu_amd.fabric import probe
from k8s_dra_driver_gpu_amd.dra import api as dra
from k8s_dra_driver_gpu_amd.daemon.process import default_fabricctl_path
import subprocess

SOAK_S = int(os.environ.get("SOAK_S", "600"))
import atexit
cluster = LocalCluster(real_devices=os.path.exists("/dev/kfd"),
                       num_gpus=2, partitionable=False,
                       work_dir=tempfile.mkdtemp(prefix="soak-")).start()
atexit.register(cluster.stop)
cluster.client.create("computedomains", {
    "apiVersion": "resource.amd.com/v1beta1", "kind": "ComputeDomain",
    "metadata": {"name": "soak-cd", "namespace": "default"},
    "spec": {"numNodes": 1}})
assert cluster.wait_cd_ready("soak-cd", "default", 60), "CD not ready"
gpu = cluster.devicelib.gpus()[0]
# SOAK_HOLD: number of PrepareCompleted claims seeded into the live
# checkpoint before churn starts. On a 1-GPU box we cannot hold real
# prepared claims on distinct devices, so the hold models the actual
# scaling concern: every churn RMW pays the cost of a large standing
# checkpoint (see the fill() below).
HOLD = int(os.environ.get("SOAK_HOLD", "0"))

def churn_one(i):
    claim = cluster.client.create("resourceclaims", {
        "apiVersion": "resource.k8s.io/v1beta1", "kind": "ResourceClaim",
        "metadata": {"name": f"soak-{uuid.uuid4().hex[:10]}", "namespace": "default"},
        "spec": {"devices": {"requests": [{"name": "r0", "deviceClassName": "gpu.amd.com"}]}}})
    cluster.scheduler.schedule_pending()
    claim = cluster.client.get("resourceclaims", claim["metadata"]["name"], "default")
    uid = claim["metadata"]["uid"]
    msg = dra.Claim(namespace="default", name=claim["metadata"]["name"], uid=uid)
    t0 = time.monotonic()
    r = cluster.gpu_client.prepare([msg]).claims[uid]
    lat = time.monotonic() - t0
    assert r.error == "", r.error
    cluster.gpu_client.unprepare([msg])
    cluster.client.delete("resourceclaims", claim["metadata"]["name"], "default")
    cluster.scheduler.release(claim)
    return lat

t_start = time.monotonic()
lats, cycles, probe_reads, errors = [], 0, [], 0
# build up a standing population of prepared CD-channel claims? channel-0 is
# exclusive; instead hold GPU claims each on its own simulated node
holders = []
if HOLD:
    import threading as _t
    from k8s_dra_driver_gpu_amd.plugin.device_state import AllocatedClaim, AllocatedDevice
    from k8s_dra_driver_gpu_amd.plugin.checkpoint import CheckpointManager, ClaimRef
    from k8s_dra_driver_gpu_amd.cdi.spec import CdiHandler
    from k8s_dra_driver_gpu_amd.plugin.device_state import DeviceState
    hd = tempfile.mkdtemp(prefix="soak-hold-")
    hstate = DeviceState(
        devicelib=cluster.devicelib,
        cdi=CdiHandler(cdi_root=os.path.join(hd, "cdi")),
        checkpoints=CheckpointManager(os.path.join(hd, "state")),
        state_dir=os.path.join(hd, "state"),
    )
    from k8s_dra_driver_gpu_amd.plugin.checkpoint import PreparedClaim, PreparedDevice, PREPARE_COMPLETED
    def fill(data):
        for i in range(HOLD):
            u = f"{i:08d}-hold-4000-8000-000000000000"
            data.set_claim(u, PreparedClaim(
                state=PREPARE_COMPLETED,
                claim=ClaimRef("soak", f"held-{i}", u),
                devices=[PreparedDevice(type="gpu", name=f"gpu-held-{i}")]))
    cluster.gpu_driver.state.checkpoints.update(fill)
    print(f"holding {HOLD} completed claims in the live checkpoint")
minute = 0
while time.monotonic() - t_start < SOAK_S:
    try:
        lats.append(churn_one(cycles)); cycles += 1
    except Exception as e:
        errors += 1
        print("churn error:", e)
    if cycles % 200 == 0:
        if os.path.exists("/dev/kfd"):
            probe_reads.append(probe.hbm_read_gbps(0, 1 << 30, 2))
        st = subprocess.run([default_fabricctl_path(), "-q", "-p",
                             str(list(cluster.supervisors.values())[0].command_port)],
                            capture_output=True, text=True, timeout=10).stdout.strip()
        el = time.monotonic() - t_start
        if el // 60 > minute:
            minute = int(el // 60)
            print(f"[{el:5.0f}s] cycles={cycles} p50={statistics.median(lats)*1e3:.2f}ms "
                  f"hbm={probe_reads[-1]:.0f}GB/s fabricd={st} errors={errors}")
        assert st.startswith("READY"), f"fabricd degraded: {st}"
cd = cluster.client.get("computedomains", "soak-cd", "default")
lats.sort()
print(json.dumps({
    "soak_seconds": round(time.monotonic() - t_start, 1),
    "cycles": cycles, "errors": errors,
    "p50_ms": round(lats[len(lats)//2]*1e3, 3),
    "p99_ms": round(lats[int(len(lats)*0.99)]*1e3, 3),
    "max_ms": round(lats[-1]*1e3, 3),
    "hbm_min_gbps": round(min(probe_reads), 1) if probe_reads else None,
    "held_claims": HOLD,
    "hbm_max_gbps": round(max(probe_reads), 1) if probe_reads else None,
    "cd_status_at_end": (cd.get("status") or {}).get("status"),
}))
cluster.stop()
PYEOF
echo "soak done"
