#!/usr/bin/env bash
# Scripted install for a real Kubernetes cluster — the analog of the
# reference's scripts/install-fma.sh + docs/install-scripted.md: the
# Helm chart covers every NAMESPACE-scoped object; cluster-scoped
# categories (CRDs, ValidatingAdmissionPolicies+bindings, the
# node-reading ClusterRole) are each behind an opt-in flag because (a)
# they cannot be owned by two users' chart instances and (b) creating
# them needs more authorization than namespace admins usually have
# (reference docs/install-scripted.md:17-48, docs/cluster-sharing.md).
#
#   deploy/install.sh -n my-ns --crds --admission-policies --cluster-rbac \
#       --registry my-registry --tag v0.2
set -euo pipefail

here="$(cd "$(dirname "$0")" && pwd)"
ns=default registry=localhost tag=latest
crds=0 vaps=0 crbac=0 dry=0

while [[ $# -gt 0 ]]; do
  case "$1" in
    -n|--namespace) ns="$2"; shift 2 ;;
    --registry) registry="$2"; shift 2 ;;
    --tag) tag="$2"; shift 2 ;;
    --crds) crds=1; shift ;;
    --admission-policies) vaps=1; shift ;;
    --cluster-rbac) crbac=1; shift ;;
    --dry-run) dry=1; shift ;;
    *) echo "unknown flag $1" >&2; exit 2 ;;
  esac
done

apply() { if [[ $dry == 1 ]]; then kubectl apply --dry-run=client -f "$1"; else kubectl apply -f "$1"; fi }

# cluster-scoped categories, each explicitly opted into
if [[ $crds == 1 ]]; then
  echo "== installing CRDs (cluster-scoped)"
  apply "$here/../manifests/crds"
fi
if [[ $vaps == 1 ]]; then
  echo "== installing ValidatingAdmissionPolicies (cluster-scoped)"
  apply "$here/../manifests/validating-admission-policies"
fi
if [[ $crbac == 1 ]]; then
  echo "== installing node-reading ClusterRole/Binding (cluster-scoped)"
  apply "$here/../manifests/kubernetes"
fi

# namespace-scoped: the chart (controllers, SA, namespaced RBAC)
echo "== installing chart into namespace $ns"
helm_args=(upgrade --install fma-amd "$here/charts/fma-amd"
           --namespace "$ns" --create-namespace
           --set "namespace=$ns"
           --set "image.registry=$registry" --set "image.tag=$tag")
if [[ $dry == 1 ]]; then helm_args+=(--dry-run); fi
helm "${helm_args[@]}"

echo "== done; populate the gpu-map with tools/ensure_nodes_mapped.py" \
     "on each GPU node (direct actuation path only)"
