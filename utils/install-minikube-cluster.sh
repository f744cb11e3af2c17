#!/bin/bash
# Single-node minikube with the AMD GPU device plugin (reference
# utils/install-minikube-cluster.sh adapted from the NVIDIA operator to
# the AMD stack: amdgpu device plugin + node labeller).
set -euo pipefail
curl -LO https://storage.googleapis.com/minikube/releases/latest/minikube-linux-amd64
sudo install minikube-linux-amd64 /usr/local/bin/minikube
minikube start --driver=docker --container-runtime=docker \
  --mount --mount-string /dev/kfd:/dev/kfd --mount-string /dev/dri:/dev/dri
bash "$(dirname "$0")/init-amd-gpu-setup-k8s.sh"
