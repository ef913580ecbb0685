#!/usr/bin/env python3
"""Multi-process soak over HTTP (run on a GPU box): mini API server +
scheduler-stub process + GPU kubelet-plugin process (real device layer),
webhook admission on every claim, churn for SOAK_S seconds.

Exercises the paths the in-process soak does not: HTTP watch streams,
informers over HTTP, cross-process scheduling, the plugin's periodic slice
refresh, and the admission webhook."""

import json
import os
import statistics
import subprocess
import sys
import tempfile
import time
import uuid

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import yaml  # noqa: E402

from k8s_dra_driver_gpu_amd.dra import api as dra  # noqa: E402
from k8s_dra_driver_gpu_amd.k8s.client import HttpClient  # noqa: E402
from k8s_dra_driver_gpu_amd.k8s.httpserver import MiniApiServer  # noqa: E402
from k8s_dra_driver_gpu_amd.webhook.server import validate_admission_review  # noqa: E402

SOAK_S = int(os.environ.get("SOAK_S", "600"))


def main() -> int:
    srv = MiniApiServer()
    srv.start()
    client = HttpClient(base_url=f"http://127.0.0.1:{srv.port}", qps=10000, burst=10000)
    from k8s_dra_driver_gpu_amd.utils.helmlite import chart_deviceclasses

    chart = os.path.join(REPO, "deployments", "helm", "amd-dra-driver")
    for doc in chart_deviceclasses(chart):
        client.create("deviceclasses", doc)

    work = tempfile.mkdtemp(prefix="httpsoak-")
    env = dict(os.environ)
    if not os.path.exists("/dev/kfd"):  # CPU dry-run: mock device layer
        from k8s_dra_driver_gpu_amd.device.mock import MockTree

        tree = MockTree(root=os.path.join(work, "mock"), num_gpus=2)
        tree.setup()
        env["AMDDRA_SYSFS_ROOT"] = tree.sysfs_root
        env["AMDDRA_DEV_ROOT"] = tree.dev_root
    env.update({
        "PYTHONPATH": REPO,
        "AMDDRA_API_SERVER": f"http://127.0.0.1:{srv.port}",
        "PLUGIN_DIR": os.path.join(work, "plugin"),
        "PLUGINS_REGISTRY_DIR": os.path.join(work, "registry"),
        "CDI_ROOT": os.path.join(work, "cdi"),
        "NODE_NAME": "soak-node",
        "SCHED_POLL_INTERVAL": "0.05",
        "AMDDRA_KUBE_QPS": "2000",
        "AMDDRA_KUBE_BURST": "2000",
    })
    logdir = os.environ.get("SOAK_LOG_DIR", "")
    def _sink(name):
        if logdir:
            os.makedirs(logdir, exist_ok=True)
            return open(os.path.join(logdir, name), "w")
        return subprocess.DEVNULL
    plugin = subprocess.Popen(
        [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.gpu_kubelet_plugin"],
        env=env, cwd=REPO, stdout=_sink("plugin.log"), stderr=subprocess.STDOUT,
    )
    sched = subprocess.Popen(
        [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.scheduler"],
        env=env, cwd=REPO, stdout=_sink("sched.log"), stderr=subprocess.STDOUT,
    )
    kubelet = None
    ok = True
    try:
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline and not client.list("resourceslices"):
            time.sleep(0.2)
        assert client.list("resourceslices"), "no slices published"
        kubelet = dra.DRAPluginClient(f"unix://{os.path.join(work, 'plugin', 'dra.sock')}")

        lats, cycles, errors = [], 0, 0
        t0 = time.monotonic()
        minute = 0
        while time.monotonic() - t0 < SOAK_S:
            name = f"s-{uuid.uuid4().hex[:10]}"
            body = {
                "apiVersion": "resource.k8s.io/v1beta1",
                "kind": "ResourceClaim",
                "metadata": {"name": name, "namespace": "default"},
                "spec": {"devices": {"requests": [
                    {"name": "r0", "deviceClassName": "gpu.amd.com"}]}},
            }
            # webhook admission exactly as the API server would invoke it
            review = {"request": {"uid": "x", "kind": {
                "group": "resource.k8s.io", "version": "v1beta1",
                "kind": "ResourceClaim"}, "object": body}}
            if not validate_admission_review(review)["response"]["allowed"]:
                errors += 1
                continue
            t1 = time.monotonic()
            claim = client.create("resourceclaims", body)
            uid = claim["metadata"]["uid"]
            # wait for the scheduler PROCESS to allocate
            timed_out = False
            while True:
                claim = client.get("resourceclaims", name, "default")
                if (claim.get("status") or {}).get("allocation"):
                    break
                if time.monotonic() - t1 > 30:
                    # count it (a sustained stall will fail the soak via the
                    # error count) but keep soaking — a single slow cycle on
                    # a contended CI box should not abort a long run
                    timed_out = True
                    break
                time.sleep(0.002)
            if timed_out:
                errors += 1
                print(f"[warn] allocation >30s for {name} "
                      f"(scheduler alive={sched.poll() is None})")
                client.delete("resourceclaims", name, "default")
                continue
            msg = dra.Claim(namespace="default", name=name, uid=uid)
            r = kubelet.prepare([msg]).claims[uid]
            if r.error:
                errors += 1
            else:
                kubelet.unprepare([msg])
            client.delete("resourceclaims", name, "default")
            lats.append(time.monotonic() - t1)
            cycles += 1
            el = time.monotonic() - t0
            if el // 60 > minute:
                minute = int(el // 60)
                print(f"[{el:5.0f}s] cycles={cycles} "
                      f"p50={statistics.median(lats)*1e3:.1f}ms errors={errors}")
        lats.sort()
        out = {
            "soak_seconds": round(time.monotonic() - t0, 1),
            "cycles": cycles, "errors": errors,
            "e2e_p50_ms": round(lats[len(lats)//2]*1e3, 2),
            "e2e_p99_ms": round(lats[int(len(lats)*0.99)]*1e3, 2),
            "plugin_alive": plugin.poll() is None,
            "scheduler_alive": sched.poll() is None,
        }
        print(json.dumps(out))
        ok = errors == 0 and cycles > 0 and out["plugin_alive"] and out["scheduler_alive"]
    finally:
        if kubelet:
            kubelet.close()
        for p in (plugin, sched):
            p.terminate()
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        srv.stop()
    print("HTTPSOAK", "PASS" if ok else "FAIL")
    return 0 if ok else 1


if __name__ == "__main__":
    raise SystemExit(main())
