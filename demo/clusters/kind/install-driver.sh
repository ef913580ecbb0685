#!/usr/bin/env bash
# Install the chart into the current kubectl context (kind or real).
set -euo pipefail
CURRENT_DIR="$(cd -- "$(dirname -- "${BASH_SOURCE[0]}")" &>/dev/null && pwd)"
source "${CURRENT_DIR}/scripts/common.sh"
require kubectl

EXTRA_ARGS=()
if [ "${MOCK_GPUS:-0}" = "1" ]; then
  EXTRA_ARGS+=(--set altSysfsRoot=/var/lib/amddra-mock/sys
               --set altDevRoot=/var/lib/amddra-mock/dev)
fi

if command -v helm >/dev/null 2>&1; then
  helm upgrade --install amd-dra-driver \
    "${REPO_ROOT}/deployments/helm/amd-dra-driver" \
    --namespace "${DRIVER_NAMESPACE}" --create-namespace \
    --set "image.repository=${DRIVER_IMAGE%%:*}" \
    --set "image.tag=${DRIVER_IMAGE##*:}" \
    "${EXTRA_ARGS[@]}"
else
  # no helm on the host: render with the in-repo helmlite subset renderer
  echo "helm not found; rendering with helmlite"
  kubectl create namespace "${DRIVER_NAMESPACE}" --dry-run=client -o yaml | kubectl apply -f -
  python3 - <<EOF | kubectl apply -n "${DRIVER_NAMESPACE}" -f -
import sys
sys.path.insert(0, "${REPO_ROOT}")
from k8s_dra_driver_gpu_amd.utils.helmlite import render_chart
out = render_chart("${REPO_ROOT}/deployments/helm/amd-dra-driver",
                   namespace="${DRIVER_NAMESPACE}")
print("\n---\n".join(out.values()))
EOF
  kubectl apply -f "${REPO_ROOT}/deployments/helm/amd-dra-driver/crds/"
fi
