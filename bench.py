#!/usr/bin/env python3
"""Flagship benchmark: synthetic ResourceClaim churn through the full DRA
prepare/unprepare pipeline, one rank per GPU (BASELINE.json metric:
"ResourceClaim p50 alloc latency + pods/sec; 8-GPU ComputeDomain bring-up
time").

Each *step* is one pod's claim lifecycle: build a random claim spec ->
NodePrepare (checkpoint 2-phase commit under flock + device-layer touch + CDI
spec write) -> NodeUnprepare (device reset + spec removal + checkpoint
removal) against this rank's physical GPU via the native sysfs device layer.

Fabric validation (BASELINE.json config 5) runs before the timed region:
each rank executes the hand-written CDNA4 HBM probe kernel on its GPU, and
for world_size > 1 an RCCL all-reduce over xGMI validates the mesh; the CD
bring-up time for the world is measured and reported in `config`.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
(the driver launches N>1 via torch.distributed.run, one rank per GPU).
"""

from __future__ import annotations

import argparse
import json
import os
import shutil
import sys
import tempfile
import time
import uuid as uuidlib

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    import socket as _sock

    sk = _sock.socket()
    sk.bind(("127.0.0.1", 0))
    p = sk.getsockname()[1]
    sk.close()
    return p


def _measure_mesh_bringup(n: int) -> float:
    """Controller + n daemon supervisors (each a real C++ fabricd) meshing
    over localhost; returns seconds from ComputeDomain creation to status
    Ready with all n nodes Ready."""
    import tempfile
    import threading

    from k8s_dra_driver_gpu_amd.controller.computedomain import (
        ComputeDomainController,
    )
    from k8s_dra_driver_gpu_amd.daemon.main import DaemonSupervisor
    from k8s_dra_driver_gpu_amd.daemon.process import default_fabricd_path
    from k8s_dra_driver_gpu_amd.k8s.client import FakeClient

    client = FakeClient()
    ctrl = ComputeDomainController(
        client, status_sync_period=0.1, cleanup_period=3600, max_nodes=max(8, n)
    ).start()
    work = tempfile.mkdtemp(prefix="amddra-mesh-")
    sups = []
    nodes = []
    try:
        t0 = time.monotonic()
        cd = client.create(
            "computedomains",
            {"apiVersion": "resource.amd.com/v1beta1", "kind": "ComputeDomain",
             "metadata": {"name": "bench-mesh", "namespace": "default"},
             "spec": {"numNodes": n}},
        )
        uid = cd["metadata"]["uid"]
        for i in range(n):
            sup = DaemonSupervisor(
                client=client, cd_uid=uid, node_name=f"n{i}",
                pod_ip="127.0.0.1", work_dir=f"{work}/f{i}", clique_id="h.0",
                peer_port=_free_port(), command_port=_free_port(),
                fabricd_path=default_fabricd_path(),
            )
            sups.append(sup)
            threading.Thread(
                target=lambda sp=sup: sp.run(ready_poll_interval=0.3), daemon=True
            ).start()
        deadline = time.monotonic() + 120
        while time.monotonic() < deadline:
            obj = client.get("computedomains", "bench-mesh", "default")
            st = obj.get("status") or {}
            nodes = st.get("nodes") or []
            if (st.get("status") == "Ready" and len(nodes) == n
                    and all(x.get("status") == "Ready" for x in nodes)):
                return round(time.monotonic() - t0, 3)
            time.sleep(0.05)
        raise RuntimeError(f"mesh domain not Ready within 120s (nodes={len(nodes)})")
    finally:
        for sup in sups:
            try:
                sup.stop()
            except Exception:  # noqa: BLE001
                pass
        ctrl.stop()


def _concurrent_process_churn(workers, steps, work_dir, rank, device_name,
                              tree_env):
    """P subprocess workers, each a standalone driver node doing
    prepare/unprepare cycles; returns the combined latency list."""
    import json as _json
    import subprocess
    import sys as _sys

    per = max(1, steps // workers)
    code = """
import json, os, sys, time, uuid
sys.path.insert(0, {repo!r})
from k8s_dra_driver_gpu_amd.cdi.spec import CdiHandler
from k8s_dra_driver_gpu_amd.device.devicelib import DeviceLib
from k8s_dra_driver_gpu_amd.dra import api as dra
from k8s_dra_driver_gpu_amd.plugin.checkpoint import CheckpointManager
from k8s_dra_driver_gpu_amd.plugin.device_state import DeviceState
from k8s_dra_driver_gpu_amd.plugin.driver import (
    GpuDriver, static_claim_resolver, AllocatedClaim, AllocatedDevice, ClaimRef)

w = sys.argv[1]
base = os.path.join({work!r}, "cw" + w)
store = {{}}
ds = DeviceState(
    devicelib=DeviceLib(),
    cdi=CdiHandler(cdi_root=os.path.join(base, "cdi")),
    checkpoints=CheckpointManager(os.path.join(base, "state")),
    state_dir=os.path.join(base, "state"))
drv = GpuDriver(state=ds, claim_resolver=static_claim_resolver(store),
                node_name="bench-node-{rank}-cw" + w)
socks = drv.start(plugin_dir=os.path.join(base, "plugin"))
cli = dra.DRAPluginClient("unix://" + socks["dra"])
lats = []
for i in range({per!r}):
    uid = str(uuid.uuid4())
    store[uid] = AllocatedClaim(
        ref=ClaimRef(namespace="bench", name="p" + str(i), uid=uid),
        devices=[AllocatedDevice(device={dev!r}, configs=[])])
    t0 = time.monotonic()
    r = cli.prepare([dra.Claim(namespace="bench", name="p" + str(i), uid=uid)])
    lats.append(time.monotonic() - t0)
    assert not r.claims[uid].error, r.claims[uid].error
    cli.unprepare([dra.Claim(uid=uid)])
    del store[uid]
cli.close()
drv.stop(grace=0.1)
print(json.dumps(lats))
"""
    repo = os.path.dirname(os.path.abspath(__file__))
    src = code.format(repo=repo, work=work_dir, rank=rank, per=per,
                      dev=device_name)
    env = dict(os.environ)
    env.update({k: v for k, v in tree_env.items() if v})
    env["PYTHONPATH"] = repo
    procs = [
        subprocess.Popen([_sys.executable, "-c", src, str(w)],
                         stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                         text=True, env=env)
        for w in range(workers)
    ]
    latencies = []
    for p in procs:
        out, err = p.communicate(timeout=600)
        if p.returncode != 0:
            raise RuntimeError(f"concurrency worker failed: {err[-800:]}")
        latencies.extend(_json.loads(out.strip().splitlines()[-1]))
    return latencies


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=int(os.environ.get("WORLD_SIZE", "1")))
    ap.add_argument("--steps", type=int, default=1000)
    ap.add_argument("--warmup", type=int, default=100)
    ap.add_argument("--concurrency", type=int, default=1,
                    help="experimental: parallel claim workers per rank, each "
                    "with its own plugin instance; single-process threading is "
                    "GIL-bound, so this under-reports true multi-node scaling. "
                    "1 = serial (the contract default)")
    args = ap.parse_args()

    import torch
    import torch.distributed as dist

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    have_cuda = torch.cuda.is_available()

    t_bringup0 = time.monotonic()
    if world_size > 1:
        backend = "nccl" if have_cuda else "gloo"
        dist.init_process_group(backend=backend)
        if have_cuda:
            torch.cuda.set_device(local_rank)

    # ------------------------------------------------------------------
    # Device layer: real sysfs on a GPU box; mock tree otherwise (CPU dev).
    # ------------------------------------------------------------------
    from k8s_dra_driver_gpu_amd.device.devicelib import DeviceLib
    from k8s_dra_driver_gpu_amd.device.mock import MockTree

    mock_root = None
    real = os.path.exists("/sys/class/kfd/kfd/topology/nodes") and have_cuda
    if real:
        lib = DeviceLib()
        gpus = lib.gpus()
        data_kind = "synthetic"
    else:
        mock_root = tempfile.mkdtemp(prefix=f"amddra-bench-{rank}-")
        tree = MockTree(root=mock_root, num_gpus=max(world_size, 1))
        tree.setup()
        lib = DeviceLib(backend=tree.backend())
        gpus = lib.gpus()
        data_kind = "synthetic-mock"
    if not gpus:
        print(json.dumps({"error": "no GPUs enumerated"}))
        sys.exit(1)
    my_gpu = gpus[local_rank % len(gpus)]

    # ------------------------------------------------------------------
    # Fabric validation (untimed): native HIP probe + RCCL all-reduce.
    # ------------------------------------------------------------------
    fabric = {}
    if have_cuda:
        from k8s_dra_driver_gpu_amd.fabric import probe

        gbps = probe.hbm_read_gbps(local_rank % max(1, probe.device_count()), 1 << 30, 3)
        fabric["hbm_read_gbps"] = round(gbps, 1)
        if rank == 0:
            # MX quantized-GEMM floors (real per-block E8M0 scales through
            # the mfma scale operands) — the production low-precision path
            try:
                fabric["gemm_fp8_mx_tflops"] = round(
                    probe.gemm_fp8_scaled_tflops(0, 4096, 3), 1)
                fabric["gemm_fp4_mx_tflops"] = round(
                    probe.gemm_fp4_scaled_tflops(0, 4096, 3), 1)
            except Exception:
                pass
        if rank == 0 and probe.device_count() > 1:
            # one xGMI link-pair measurement for the record (per-link ~153 GB/s
            # x links between the pair; full matrix via fabric.probe)
            try:
                fabric["xgmi_p2p_gbps"] = round(probe.p2p_read_gbps(0, 1, 256 << 20, 3), 1)
            except Exception:
                fabric["xgmi_p2p_gbps"] = -1.0
    if world_size > 1:
        dev = torch.device("cuda", local_rank) if have_cuda else torch.device("cpu")
        x = torch.ones(64 << 20 if have_cuda else 1 << 10, dtype=torch.float32, device=dev)
        t0 = time.monotonic()
        dist.all_reduce(x)
        if have_cuda:
            torch.cuda.synchronize()
        fabric["rccl_allreduce_ok"] = bool(x[0].item() == world_size)
        fabric["rccl_allreduce_s"] = round(time.monotonic() - t0, 4)
    cd_bringup_s = time.monotonic() - t_bringup0
    # Real ComputeDomain bring-up (BASELINE "8-GPU ComputeDomain bring-up
    # time"): rank 0 runs the actual controller + clique machinery + C++
    # fabricd for a single-node domain over this box's GPUs and times
    # creation -> Ready. Untimed w.r.t. the churn metric.
    if rank == 0:
        try:
            from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster

            cl = LocalCluster(
                # mock needs >=2 GPUs for an xGMI hive (clique identity)
                num_gpus=max(2, world_size), real_devices=real,
                work_dir=tempfile.mkdtemp(prefix="amddra-cdbench-"),
                partitionable=False,
            ).start()
            try:
                t0 = time.monotonic()
                cl.client.create(
                    "computedomains",
                    {"apiVersion": "resource.amd.com/v1beta1", "kind": "ComputeDomain",
                     "metadata": {"name": "bench-cd", "namespace": "default"},
                     "spec": {"numNodes": 1}},
                )
                if cl.wait_cd_ready("bench-cd", "default", timeout=60.0):
                    fabric["cd_bringup_real_s"] = round(time.monotonic() - t0, 3)
                else:
                    fabric["cd_bringup_real_s"] = -1.0
            finally:
                cl.stop()
        except Exception as e:  # noqa: BLE001
            fabric["cd_bringup_error"] = str(e)[:200]
        # N-daemon domain (the BASELINE "8-GPU ComputeDomain bring-up"
        # shape at N=8): controller + world_size real fabricd daemons
        # meshing over localhost, creation -> status Ready with all
        # world_size nodes Ready.
        if world_size > 1:
            try:
                fabric["cd_mesh_bringup_s"] = _measure_mesh_bringup(world_size)
                fabric["cd_mesh_daemons"] = world_size
            except Exception as e:  # noqa: BLE001
                fabric["cd_mesh_error"] = str(e)[:200]

    # ------------------------------------------------------------------
    # The churn harness: full plugin state machine against this rank's GPU.
    # ------------------------------------------------------------------
    from k8s_dra_driver_gpu_amd.cdi.spec import CdiHandler
    from k8s_dra_driver_gpu_amd.plugin.checkpoint import CheckpointManager, ClaimRef
    from k8s_dra_driver_gpu_amd.plugin.device_state import (
        AllocatedClaim,
        AllocatedDevice,
        DeviceState,
    )

    from k8s_dra_driver_gpu_amd.dra import api as dra
    from k8s_dra_driver_gpu_amd.plugin.driver import GpuDriver

    work_dir = tempfile.mkdtemp(prefix=f"amddra-bench-state-{rank}-")

    def make_node(tag):
        """One simulated node: plugin served on a unix socket + fake kubelet
        client — each step measures the full NodePrepareResources/
        NodeUnprepareResources path."""
        state_dir = os.path.join(work_dir, tag, "state")
        ds = DeviceState(
            devicelib=lib,
            cdi=CdiHandler(cdi_root=os.path.join(work_dir, tag, "cdi"),
                           dev_root=(lib.backend.dev_root)),
            checkpoints=CheckpointManager(state_dir),
            state_dir=state_dir,
        )
        store = {}
        drv = GpuDriver(
            state=ds,
            claim_resolver=lambda ns, name, uid: store[uid],
            node_name=f"bench-node-{rank}-{tag}",
        )
        socks = drv.start(plugin_dir=os.path.join(work_dir, tag, "plugin"))
        cli = dra.DRAPluginClient(f"unix://{socks['dra']}")
        return drv, cli, store

    driver, kubelet, alloc_store = make_node("w0")

    from k8s_dra_driver_gpu_amd.api.configs import APIVERSION

    cfg_pool = [
        None,
        {"apiVersion": APIVERSION, "kind": "GpuConfig"},
        {
            "apiVersion": APIVERSION,
            "kind": "GpuConfig",
            "sharing": {"strategy": "TimeSlicing", "timeSlicingConfig": {"interval": "Long"}},
        },
    ]

    def one_step(i: int) -> float:
        uid = str(uuidlib.uuid4())
        cfg = cfg_pool[i % len(cfg_pool)]
        alloc_store[uid] = AllocatedClaim(
            ref=ClaimRef(namespace="bench", name=f"pod-{i}", uid=uid),
            devices=[
                AllocatedDevice(
                    device=my_gpu.canonical_name, configs=[cfg] if cfg else []
                )
            ],
        )
        claim_msg = dra.Claim(namespace="bench", name=f"pod-{i}", uid=uid)
        t0 = time.monotonic()
        resp = kubelet.prepare([claim_msg])
        alloc_latency = time.monotonic() - t0
        if resp.claims[uid].error:
            raise RuntimeError(f"prepare failed: {resp.claims[uid].error}")
        uresp = kubelet.unprepare([claim_msg])
        if uresp.claims[uid].error:
            raise RuntimeError(f"unprepare failed: {uresp.claims[uid].error}")
        del alloc_store[uid]
        return alloc_latency

    for i in range(args.warmup):
        one_step(i)

    if world_size > 1:
        dist.barrier()
    if have_cuda:
        torch.cuda.synchronize()
    t_start = time.monotonic()
    if args.concurrency <= 1:
        latencies = [one_step(i) for i in range(args.steps)]
    else:
        # real concurrency: P worker PROCESSES, each a full driver node
        # (its own plugin, state dir and gRPC socket) churning in parallel —
        # the multiple-driver-pods-per-node shape, not GIL-bound threads
        latencies = _concurrent_process_churn(
            args.concurrency, args.steps, work_dir, rank,
            my_gpu.canonical_name,
            tree_env=(
                {} if real else
                {"AMDDRA_SYSFS_ROOT": lib.backend.sysfs_root,
                 "AMDDRA_DEV_ROOT": lib.backend.dev_root}
            ),
        )
    if have_cuda:
        torch.cuda.synchronize()
    elapsed = time.monotonic() - t_start
    if world_size > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=torch.device("cuda", local_rank) if have_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
        dist.barrier()
    if have_cuda:
        torch.cuda.synchronize()

    latencies.sort()
    p50 = latencies[len(latencies) // 2]
    p99 = latencies[min(len(latencies) - 1, int(len(latencies) * 0.99))]
    actual_steps = len(latencies)
    pods_per_sec = world_size * actual_steps / elapsed

    kubelet.close()
    driver.stop()
    shutil.rmtree(work_dir, ignore_errors=True)
    if mock_root:
        shutil.rmtree(mock_root, ignore_errors=True)

    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "resourceclaim_pods_per_sec",
                    "value": round(pods_per_sec, 2),
                    "unit": "pods/s",
                    "n_gpus": world_size,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": round(elapsed / max(1, actual_steps) * 1000, 3),
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    # control-plane metric: no compute dtype is involved in
                    # the timed region (claim churn over gRPC + checkpoint
                    # fsyncs + CDI writes)
                    "dtype": "n/a",
                    "data": data_kind,
                    "config": {
                        "model": "dra-claim-churn",
                        "global_batch": world_size * args.steps,
                        "seq_len": 1,
                        "parallelism": f"dp{world_size}",
                        "concurrency": args.concurrency,
                        "gpu": my_gpu.product_name,
                        "p50_alloc_latency_ms": round(p50 * 1000, 3),
                        "p99_alloc_latency_ms": round(p99 * 1000, 3),
                        "cd_bringup_s": round(
                            fabric.get("cd_bringup_real_s", cd_bringup_s), 3
                        ),
                        "fabric": fabric,
                    },
                }
            )
        )
    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
