#!/usr/bin/env bash
set -euo pipefail
CURRENT_DIR="$(cd -- "$(dirname -- "${BASH_SOURCE[0]}")" &>/dev/null && pwd)"
source "${CURRENT_DIR}/scripts/common.sh"
require kind
kind delete cluster --name "${KIND_CLUSTER_NAME}"
