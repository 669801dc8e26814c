#!/usr/bin/env bash
# Uninstall — the analog of the reference's
# docs/uninstall-step-by-step.md: namespace-scoped objects go with the
# chart release; cluster-scoped categories are only removed when asked,
# because other users of the shared cluster may still depend on them
# (reference docs/cluster-sharing.md).
#
#   deploy/uninstall.sh -n my-ns [--crds] [--admission-policies] [--cluster-rbac]
set -euo pipefail

here="$(cd "$(dirname "$0")" && pwd)"
ns=default crds=0 vaps=0 crbac=0

while [[ $# -gt 0 ]]; do
  case "$1" in
    -n|--namespace) ns="$2"; shift 2 ;;
    --crds) crds=1; shift ;;
    --admission-policies) vaps=1; shift ;;
    --cluster-rbac) crbac=1; shift ;;
    *) echo "unknown flag $1" >&2; exit 2 ;;
  esac
done

echo "== removing chart release from namespace $ns"
helm uninstall fma-amd --namespace "$ns" || true

if [[ $vaps == 1 ]]; then
  echo "== removing ValidatingAdmissionPolicies"
  kubectl delete --ignore-not-found -f "$here/../manifests/validating-admission-policies"
fi
if [[ $crbac == 1 ]]; then
  echo "== removing node-reading ClusterRole/Binding"
  kubectl delete --ignore-not-found -f "$here/../manifests/kubernetes"
fi
if [[ $crds == 1 ]]; then
  echo "== removing CRDs (deletes every ISC/LC/LPP in the cluster!)"
  kubectl delete --ignore-not-found -f "$here/../manifests/crds"
fi
