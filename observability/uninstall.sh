#!/bin/bash
# Tear down the kube-prometheus-stack observability tier installed by
# install.sh (capability parity: reference observability/uninstall.sh).
set -e
NAMESPACE=${1:-monitoring}
helm uninstall prom-adapter -n "$NAMESPACE" || true
helm uninstall kube-prom-stack -n "$NAMESPACE" || true
kubectl delete -f "$(dirname "$0")/prom-adapter.yaml" --ignore-not-found
kubectl delete namespace "$NAMESPACE" --ignore-not-found
