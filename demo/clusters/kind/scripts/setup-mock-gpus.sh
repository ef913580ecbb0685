#!/usr/bin/env bash
# Generate the mock MI355X device tree inside every kind worker node — the
# analog of the reference's hack/ci/mock-nvml/setup-mock-gpu.sh, except the
# mock is a re-rooted sysfs/dev tree consumed by the PRODUCTION backend
# (device/sysfs.py reads AMDDRA_SYSFS_ROOT/AMDDRA_DEV_ROOT), not a swapped
# C library: one code path for mock and real.
set -euo pipefail
CURRENT_DIR="$(cd -- "$(dirname -- "${BASH_SOURCE[0]}")" &>/dev/null && pwd)"
source "${CURRENT_DIR}/common.sh"
require kind
require docker

MOCK_ROOT=/var/lib/amddra-mock

for node in $(kind get nodes --name "${KIND_CLUSTER_NAME}" | grep -v control-plane); do
  echo "==> generating ${NUM_MOCK_GPUS}-GPU mock tree on ${node}"
  # the driver image carries the python package; use it to build the tree
  # into a hostPath shared with the kubelet-plugin DaemonSet
  docker exec "${node}" mkdir -p "${MOCK_ROOT}"
  docker run --rm -v "/var/lib/docker/volumes:/var/lib/docker/volumes" \
    --volumes-from "${node}" "${DRIVER_IMAGE}" \
    python -c "from k8s_dra_driver_gpu_amd.device.mock import MockTree; \
               MockTree('${MOCK_ROOT}', num_gpus=${NUM_MOCK_GPUS}).setup(); \
               print('mock tree ready')" || {
    # fallback: run inside the node image if the driver image isn't loaded
    docker exec "${node}" sh -c "command -v python3" >/dev/null && \
      docker cp "${REPO_ROOT}/k8s_dra_driver_gpu_amd" "${node}:/tmp/amddra-pkg/k8s_dra_driver_gpu_amd" && \
      docker exec -e PYTHONPATH=/tmp/amddra-pkg "${node}" \
        python3 -c "from k8s_dra_driver_gpu_amd.device.mock import MockTree; \
                    MockTree('${MOCK_ROOT}', num_gpus=${NUM_MOCK_GPUS}).setup(); \
                    print('mock tree ready')"
  }
done

echo "Install the chart with:"
echo "  --set altSysfsRoot=${MOCK_ROOT}/sys --set altDevRoot=${MOCK_ROOT}/dev"
