#!/usr/bin/env python3
"""Run the quickstart demo specs against the full driver stack in one
process (fake API server + mock MI355X sysfs tree + real plugins/controller/
fabricd) — the kind-cluster demo analog. Usage: python demo/run_local.py"""

import logging
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from k8s_dra_driver_gpu_amd.bench.localcluster import LocalCluster  # noqa: E402

SPECS = os.path.join(os.path.dirname(os.path.abspath(__file__)), "specs", "quickstart")


def main() -> int:
    logging.basicConfig(level=logging.WARNING)
    cluster = LocalCluster(num_gpus=16, vfio=True).start()
    try:
        for spec in ("gpu-test1.yaml", "gpu-test2.yaml", "gpu-test3.yaml",
                     "gpu-test5.yaml", "gpu-test7.yaml", "gpu-test8.yaml",
                     "gpu-test-partitions.yaml",
                     "gpu-test-vfio.yaml", "gpu-test-extres.yaml"):
            print(f"=== {spec} ===")
            for ev in cluster.apply_yaml(os.path.join(SPECS, spec)):
                print(" ", ev)
        print("=== cd-test1.yaml (ComputeDomain bring-up) ===")
        t0 = time.monotonic()
        for ev in cluster.apply_yaml(os.path.join(SPECS, "cd-test1.yaml")):
            print(" ", ev)
        ready = cluster.wait_cd_ready("cd1", "cd-test1")
        print(f"  ComputeDomain cd1 Ready={ready} in {time.monotonic()-t0:.1f}s")
        cd = cluster.client.get("computedomains", "cd1", "cd-test1")
        for n in (cd.get("status") or {}).get("nodes", []):
            print(f"    node {n['name']} idx={n['index']} clique={n['cliqueID']} {n['status']}")
        return 0 if ready else 1
    finally:
        cluster.stop()


if __name__ == "__main__":
    raise SystemExit(main())
