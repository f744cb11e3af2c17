#!/bin/bash
# Make MI355X GPUs schedulable in a cluster (reference
# utils/init-nvidia-gpu-setup-k8s.sh counterpart for AMD): deploys the
# AMD device plugin (exposes amd.com/gpu, the resource name
# helm/values.yaml uses) and the node labeller (VRAM/CU labels).
set -euo pipefail
kubectl apply -f https://raw.githubusercontent.com/ROCm/k8s-device-plugin/master/k8s-ds-amdgpu-dp.yaml
kubectl apply -f https://raw.githubusercontent.com/ROCm/k8s-device-plugin/master/k8s-ds-amdgpu-labeller.yaml
echo "waiting for amd.com/gpu to be allocatable..."
for i in $(seq 1 60); do
  if kubectl get nodes -o json | grep -q '"amd.com/gpu"'; then
    kubectl get nodes -o custom-columns='NAME:.metadata.name,GPUS:.status.allocatable.amd\.com/gpu'
    exit 0
  fi
  sleep 5
done
echo "amd.com/gpu never became allocatable" >&2
exit 1
