#!/usr/bin/env bash
# Create a kind cluster ready for the AMD DRA driver with the FAKE HAL —
# the whole driver (enumeration, ResourceSlices, prepare, partitioning,
# sharing) runs against a modeled 8xMI355X node, no GPUs required
# (BASELINE config #1; reference analog: demo/clusters/kind with real
# GPUs injected — the fake HAL replaces the nvkind device-masking trick).
set -euo pipefail

HERE="$(cd "$(dirname "$0")" && pwd)"
CLUSTER_NAME="${CLUSTER_NAME:-amd-dra}"
K8S_IMAGE="${K8S_IMAGE:-kindest/node:v1.32.0}"

kind create cluster --name "${CLUSTER_NAME}" --image "${K8S_IMAGE}" \
     --config "${HERE}/kind-cluster-config.yaml"

# DRA needs the feature gate + API (enabled via the cluster config) and
# containerd CDI injection (enabled via the containerd patch below).
echo "Cluster '${CLUSTER_NAME}' ready."
echo "Install the driver with:"
echo "  helm install amd-dra deployments/helm/k8s-dra-driver-amd \\"
echo "      --set kubeletPlugin.hal=fake \\"
echo "      --set 'kubeletPlugin.nodeSelector=null'"
