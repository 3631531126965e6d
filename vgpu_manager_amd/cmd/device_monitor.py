"""device-monitor binary (reference cmd/device-monitor): Prometheus
exporter + optional shared utilization sampler."""
from __future__ import annotations

import argparse
import logging
import os
import sys
import threading

from prometheus_client import REGISTRY

from ..monitor.collector import NodeVgpuCollector, PhysicalGpuCollector
from ..monitor.lister import ContainerLister
from ..util import consts


def main(argv=None):
    ap = argparse.ArgumentParser("vgpu-device-monitor")
    ap.add_argument("--node-name",
                    default=os.environ.get("NODE_NAME", "node"))
    ap.add_argument("--port", type=int, default=9394)
    ap.add_argument("--base-dir", default=consts.MANAGER_DIR)
    ap.add_argument("--shared-watcher", action="store_true",
                    help="run the shared utilization sampler too")
    ap.add_argument("--tls-cert", default="")
    ap.add_argument("--tls-key", default="")
    ap.add_argument("--scrape-rate", type=float, default=2.0,
                    help="max /metrics scrapes per second")
    ap.add_argument("--stuck-grace-period", type=int, default=0,
                    help="seconds after which a pre-allocated-but-"
                         "never-bound pod is recovered (0 = off)")
    ap.add_argument("--dra-checkpoint",
                    default="/var/lib/vgpu-manager/checkpoint.json",
                    help="DRA driver checkpoint to export claim "
                         "metrics from (if present)")
    args = ap.parse_args(argv)
    logging.basicConfig(level=logging.INFO)

    lister = ContainerLister(base_dir=args.base_dir)
    try:
        from ..device.manager import AmdDeviceManager
        manager = AmdDeviceManager(args.node_name)
    except Exception:
        from ..device.manager import FakeDeviceManager
        manager = FakeDeviceManager(args.node_name, n_devices=0)
    REGISTRY.register(NodeVgpuCollector(manager, lister))
    REGISTRY.register(PhysicalGpuCollector())
    from ..monitor.collector import DraClaimCollector
    REGISTRY.register(DraClaimCollector(args.dra_checkpoint,
                                        args.node_name))

    if args.shared_watcher:
        from ..monitor.sampler import AmdSmiSource, UtilSampler
        sampler = UtilSampler(
            AmdSmiSource(),
            os.path.join(args.base_dir, "watcher", "sm_util.config"))
        sampler.start_background()

    if args.stuck_grace_period > 0:
        from ..client.kube import RestKubeClient
        from ..monitor.stuck import recover_stuck_pods

        def stuck_loop():
            import time as _t
            client = RestKubeClient()
            while True:
                try:
                    recover_stuck_pods(
                        client,
                        default_grace_s=args.stuck_grace_period)
                except Exception as e:
                    logging.warning("stuck-pod pass failed: %s", e)
                _t.sleep(max(args.stuck_grace_period / 2, 30))

        threading.Thread(target=stuck_loop, daemon=True,
                         name="stuck-pods").start()

    from ..monitor.server import serve_metrics
    serve_metrics(args.port, certfile=args.tls_cert or None,
                  keyfile=args.tls_key or None,
                  rate=args.scrape_rate)
    threading.Event().wait()
    return 0


if __name__ == "__main__":
    sys.exit(main())
