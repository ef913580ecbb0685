#!/usr/bin/env bash
# Shared settings for the kind demo cluster (the analog of the reference's
# demo/clusters/kind/scripts/common.sh).

: "${KIND_CLUSTER_NAME:=amd-dra-driver-cluster}"
: "${DRIVER_IMAGE:=amd-dra-driver:latest}"
: "${DRIVER_NAMESPACE:=amd-dra-driver}"
: "${NUM_MOCK_GPUS:=8}"

SCRIPTS_DIR="$(cd -- "$(dirname -- "${BASH_SOURCE[0]}")" &>/dev/null && pwd)"
KIND_DIR="$(dirname "${SCRIPTS_DIR}")"
REPO_ROOT="$(cd "${KIND_DIR}/../../.." && pwd)"

require() {
  command -v "$1" >/dev/null 2>&1 || {
    echo "ERROR: '$1' is required but not installed" >&2
    exit 1
  }
}
