#!/usr/bin/env bash
# Create a kind cluster ready for the AMD DRA driver (DRA feature gates +
# CDI-enabled containerd), optionally seed mock MI355X GPUs, and install
# the chart. Analog of the reference's demo/clusters/kind/create-cluster.sh.
#
#   MOCK_GPUS=1 ./create-cluster.sh     # CPU-only: 8 emulated MI355X/node
#   ./create-cluster.sh                 # real MI355X hosts (device passthrough)
set -euo pipefail
CURRENT_DIR="$(cd -- "$(dirname -- "${BASH_SOURCE[0]}")" &>/dev/null && pwd)"
source "${CURRENT_DIR}/scripts/common.sh"
require kind
require kubectl

kind create cluster --name "${KIND_CLUSTER_NAME}" \
  --config "${KIND_DIR}/kind-cluster-config.yaml"

# load a locally-built driver image when present
if command -v docker >/dev/null 2>&1 &&
   [ -n "$(docker images --filter "reference=${DRIVER_IMAGE}" -q)" ]; then
  kind load docker-image --name "${KIND_CLUSTER_NAME}" "${DRIVER_IMAGE}"
fi

if [ "${MOCK_GPUS:-0}" = "1" ]; then
  "${SCRIPTS_DIR}/setup-mock-gpus.sh"
fi

"${CURRENT_DIR}/install-driver.sh"

echo
echo "Cluster '${KIND_CLUSTER_NAME}' ready. Try:"
echo "  kubectl get resourceslices"
echo "  kubectl apply -f ${REPO_ROOT}/demo/specs/quickstart/gpu-test1.yaml"
