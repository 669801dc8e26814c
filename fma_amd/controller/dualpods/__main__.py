"""dual-pods-controller entry point (reference cmd/dual-pods-controller)."""
import argparse
import time

from fma_amd.controller.dualpods.controller import (ControllerConfig,
                                                    DualPodsController)
from fma_amd.controller.httpadapter import HttpAdapter
from fma_amd.controller import metrics
from fma_amd.store.client import StoreClient

def make_store(args, actor):
    """--backend store: our single-node store server (StoreClient);
    --backend kube: a Kubernetes apiserver at --store-url (KubeStore);
    --backend in-cluster: the surrounding cluster via the Pod's
    service-account mount (what the Helm chart deploys)."""
    if args.backend == "store":
        return StoreClient(args.store_url, actor=actor), args.namespace
    from fma_amd.store.kubestore import KubeStore
    if args.backend == "kube":
        return KubeStore(args.store_url, actor=actor), args.namespace
    ks = KubeStore.in_cluster(actor=actor)
    ns = (args.namespace if args.namespace != "default"
          else KubeStore.in_cluster_namespace())
    return ks, ns


def main():
    ap = argparse.ArgumentParser("fma-dual-pods-controller")
    ap.add_argument("--store-url", default="http://127.0.0.1:8081")
    ap.add_argument("--backend", default="store",
                    choices=("store", "kube", "in-cluster"))
    ap.add_argument("--namespace", default="default")
    # flags per reference cmd/dual-pods-controller/main.go:44-81
    ap.add_argument("--sleeper-limit", type=int, default=1)
    ap.add_argument("--num-workers", type=int, default=2)
    ap.add_argument("--debug-accelerator-memory", type=int, default=None,
                    help="MiB budget per sleeping accelerator")
    ap.add_argument("--metrics-port", type=int, default=8002)
    ap.add_argument("--debug-port", type=int, default=8003)
    args = ap.parse_args()
    store, ns = make_store(args, "dual-pods-controller")
    ctl = DualPodsController(
        store, HttpAdapter(observe=metrics.observe_http),
        ControllerConfig(
            namespace=ns,
            sleeper_limit=args.sleeper_limit,
            num_workers=args.num_workers,
            accelerator_sleeping_memory_limit_mib=(
                args.debug_accelerator_memory * args.sleeper_limit
                if args.debug_accelerator_memory else None)))
    metrics.serve_metrics(args.metrics_port)
    metrics.serve_debug(args.debug_port)
    ctl.start()
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        ctl.stop()


if __name__ == "__main__":
    main()
