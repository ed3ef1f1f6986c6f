#!/usr/bin/env bash
# Create a GKE cluster with the DRA beta APIs enabled (reference analog:
# demo/clusters/gke/create-cluster.sh — unstable-API flag + node pool).
# Requires a GKE version exposing resource.k8s.io/v1beta1 and nodes with
# AMD Instinct accelerators (or use --num-nodes with the fake HAL for a
# control-plane-only demo).
set -euo pipefail

CLUSTER_NAME="${CLUSTER_NAME:-amd-dra-demo}"
ZONE="${ZONE:-us-central1-a}"
VERSION="${GKE_VERSION:-1.32}"

gcloud container clusters create "${CLUSTER_NAME}" \
  --zone "${ZONE}" \
  --cluster-version "${VERSION}" \
  --enable-kubernetes-unstable-apis=resource.k8s.io/v1beta1/deviceclasses,resource.k8s.io/v1beta1/resourceclaims,resource.k8s.io/v1beta1/resourceclaimtemplates,resource.k8s.io/v1beta1/resourceslices \
  --no-enable-autorepair \
  --no-enable-autoupgrade \
  --num-nodes 1

echo "Install the driver:"
echo "  helm install amd-dra deployments/helm/k8s-dra-driver-amd"
