"""launcher-populator entry point (reference cmd/launcher-populator)."""
import argparse
import time

from fma_amd.controller import metrics
from fma_amd.controller.populator.populator import LauncherPopulator
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
    ap = argparse.ArgumentParser("fma-launcher-populator")
    ap.add_argument("--store-url", default="http://127.0.0.1:8081")
    ap.add_argument("--backend", default="store",
                    choices=("store", "kube", "in-cluster"))
    ap.add_argument("--namespace", default="default")
    ap.add_argument("--key-workers", type=int, default=4)
    ap.add_argument("--metrics-port", type=int, default=8004)
    ap.add_argument("--debug-port", type=int, default=8005)
    args = ap.parse_args()
    store, ns = make_store(args, "launcher-populator")
    pop = LauncherPopulator(store, namespace=ns,
                            key_workers=args.key_workers)
    metrics.serve_metrics(args.metrics_port)
    metrics.serve_debug(args.debug_port)
    pop.start()
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        pop.stop()


if __name__ == "__main__":
    main()
