#!/usr/bin/env bash
# WVA-AMD installer (reference deploy/install.sh analog).
#
#   ./deploy/install.sh [kubernetes|openshift] [NAMESPACE]
#
# Applies the CRD, ConfigMaps and the Helm chart. INFRA_ONLY=1 skips the
# manager deployment (CRD + config only).
set -euo pipefail

ENV="${1:-kubernetes}"
NS="${2:-wva-system}"
HERE="$(cd "$(dirname "$0")" && pwd)"

echo ">>> environment=${ENV} namespace=${NS}"
kubectl get ns "${NS}" >/dev/null 2>&1 || kubectl create ns "${NS}"

echo ">>> applying VariantAutoscaling CRD"
kubectl apply -f "${HERE}/crd/llmd.ai_variantautoscalings.yaml"

echo ">>> applying ConfigMaps"
kubectl apply -n "${NS}" -f "${HERE}/manager/configmaps.yaml"

if [ "${INFRA_ONLY:-0}" = "1" ]; then
  echo ">>> INFRA_ONLY=1: skipping manager deployment"
  exit 0
fi

echo ">>> installing chart"
EXTRA=""
if [ "${ENV}" = "openshift" ]; then
  EXTRA="--set openshift=true"
fi
helm upgrade --install wva-amd "${HERE}/chart/wva-amd" \
  -n "${NS}" ${EXTRA} \
  --set prometheus.baseUrl="${PROMETHEUS_BASE_URL:-http://prometheus-k8s.monitoring:9090}"

echo ">>> done; verify with: kubectl -n ${NS} get pods && kubectl get va -A"
