"""launcher-populator entry point (reference cmd/launcher-populator)."""
import argparse
import time

from fma_amd.controller import metrics
from fma_amd.controller.populator.populator import LauncherPopulator
from fma_amd.store.client import StoreClient


def main():
    ap = argparse.ArgumentParser("fma-launcher-populator")
    ap.add_argument("--store-url", default="http://127.0.0.1:8081")
    ap.add_argument("--namespace", default="default")
    ap.add_argument("--key-workers", type=int, default=4)
    ap.add_argument("--metrics-port", type=int, default=8004)
    ap.add_argument("--debug-port", type=int, default=8005)
    args = ap.parse_args()
    pop = LauncherPopulator(StoreClient(args.store_url,
                                        actor="launcher-populator"),
                            namespace=args.namespace,
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
