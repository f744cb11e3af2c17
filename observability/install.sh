#!/bin/bash
# Install kube-prometheus-stack wired for the production-stack-amd metrics
# (reference observability/install.sh parity).
set -euo pipefail
NS="${1:-monitoring}"
helm repo add prometheus-community https://prometheus-community.github.io/helm-charts || true
helm repo update
helm upgrade --install kube-prom-stack prometheus-community/kube-prometheus-stack \
  --namespace "$NS" --create-namespace \
  -f "$(dirname "$0")/kube-prom-stack.yaml"
kubectl apply -f "$(dirname "$0")/prom-adapter.yaml" -n "$NS" || true
echo "Grafana: kubectl port-forward -n $NS svc/kube-prom-stack-grafana 3000:80"
