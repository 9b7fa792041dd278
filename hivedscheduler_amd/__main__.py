"""Entry point: `python -m hivedscheduler_amd [--config PATH] [--standalone]`.

Parity with reference cmd/hivedscheduler/main.go + scheduler Run()
(scheduler.go:196-216): load config, watch it (exit(0) on change for
work-preserving reconfiguration), start node/pod informers, wait for cache
sync (recovery of bound pods), then serve the extender HTTP API.

--standalone runs without a K8s API server (all config nodes marked healthy;
useful for local evaluation and the simulated BASELINE configs).
"""
from __future__ import annotations

import argparse
import logging
import os
import sys


def main() -> None:
    ap = argparse.ArgumentParser(prog="hivedscheduler-amd")
    ap.add_argument("--config", default=None, help="config YAML (default: $CONFIG)")
    ap.add_argument("--standalone", action="store_true",
                    help="run without a K8s API server (all nodes healthy)")
    ap.add_argument("--address", default=None, help="override webServerAddress")
    ap.add_argument("-v", "--verbose", action="store_true")
    args = ap.parse_args()

    logging.basicConfig(
        level=logging.DEBUG if args.verbose else logging.INFO,
        format="%(asctime)s %(levelname).1s %(name)s: %(message)s",
    )
    log = logging.getLogger("hivedscheduler")

    from .api import config as apicfg
    from .scheduler import HivedScheduler
    from .webserver import run_server

    cfg_path = args.config or os.environ.get("CONFIG")
    if cfg_path:
        config = apicfg.init_raw_config(cfg_path)
        apicfg.watch_config(cfg_path)
    else:
        log.error("no config: pass --config or set $CONFIG")
        sys.exit(1)

    k8s = None
    if not args.standalone:
        from .k8s import Informer, KubeClient

        try:
            k8s = KubeClient(api_server=config.kubeApiServerAddress)
        except ValueError as e:
            log.error("cannot reach K8s (%s); use --standalone for local runs", e)
            sys.exit(1)

    scheduler = HivedScheduler(config, k8s_client=k8s)

    if args.standalone:
        for node in scheduler.algorithm.all_nodes():
            scheduler.algorithm.set_healthy_node(node)
        log.info("standalone mode: %d nodes marked healthy",
                 len(scheduler.algorithm.all_nodes()))
    else:
        from .k8s import Informer

        node_informer = Informer(
            k8s, "/api/v1/nodes",
            on_add=scheduler.on_node_add,
            on_update=scheduler.on_node_update,
            on_delete=scheduler.on_node_delete,
        ).start()
        pod_informer = Informer(
            k8s, "/api/v1/pods",
            on_add=scheduler.on_pod_add,
            on_update=scheduler.on_pod_update,
            on_delete=scheduler.on_pod_delete,
        ).start()
        log.info("waiting for informer cache sync (recovery of bound pods)...")
        if not (node_informer.wait_for_cache_sync(120) and pod_informer.wait_for_cache_sync(120)):
            log.error("informer cache sync timed out")
            sys.exit(1)
        log.info("cache synced: %d nodes, %d pods",
                 len(node_informer.cache), len(pod_informer.cache))

    scheduler.synced.set()
    run_server(scheduler, address=args.address)


if __name__ == "__main__":
    main()
