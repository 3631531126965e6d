"""kubelet-plugin binary (reference cmd/kubelet-plugin): the DRA
driver — mutually exclusive with the device plugin on a node.
Serves DRAPlugin + Registration sockets, publishes ResourceSlices,
optionally answers NRI-style container adjustments (dra/nri.py).
"""
from __future__ import annotations

import argparse
import logging
import os
import signal
import sys
import threading

from ..client.kube import RestKubeClient
from ..device.manager import AmdDeviceManager
from ..dra import api
from ..dra.driver import DraDriver, DraDriverServer, default_endpoint
from ..dra.state import DeviceState
from ..util import consts
from ..util.nodeconfig import DRA_GATES, FeatureGates, load_node_config


def main(argv=None):
    ap = argparse.ArgumentParser("vgpu-kubelet-plugin")
    ap.add_argument("--node-name",
                    default=os.environ.get("NODE_NAME", ""))
    ap.add_argument("--node-config-path", default=None)
    ap.add_argument("--feature-gates", default="")
    ap.add_argument("--claims-dir",
                    default="/var/lib/vgpu-manager/claims")
    ap.add_argument("--checkpoint",
                    default="/var/lib/vgpu-manager/checkpoint.json")
    ap.add_argument("--plugins-dir", default=api.PLUGINS_DIR)
    ap.add_argument("--plugins-registry", default=api.PLUGINS_REGISTRY)
    ap.add_argument("--domain", default=consts.AMD_DOMAIN)
    args = ap.parse_args(argv)

    logging.basicConfig(level=logging.INFO)
    consts.set_domain(args.domain)
    if not args.node_name:
        ap.error("--node-name or NODE_NAME required")

    gates = FeatureGates(DRA_GATES)
    gates.parse(args.feature_gates)
    gates.validate(gates.as_dict())

    config = load_node_config(args.node_config_path, args.node_name)
    client = RestKubeClient()
    manager = AmdDeviceManager(args.node_name, config)

    pm = None
    if gates.enabled("DynamicCPXPartitioning"):
        from ..device.partition import (
            AmdSmiPartitionBackend,
            PartitionManager,
        )
        pm = PartitionManager(AmdSmiPartitionBackend())
    state = DeviceState(args.node_name, manager.devices,
                        claims_dir=args.claims_dir,
                        checkpoint_path=args.checkpoint,
                        partition_manager=pm)
    driver = DraDriver(state, client,
                       endpoint=default_endpoint(args.plugins_dir))
    server = DraDriverServer(driver, plugins_dir=args.plugins_dir,
                             plugins_registry=args.plugins_registry)
    server.start()
    driver.publish_resource_slices(
        consumable_shares=gates.enabled("ConsumableShares"),
        cpx=gates.enabled("DynamicCPXPartitioning"))

    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    signal.signal(signal.SIGINT, lambda *_: stop.set())
    stop.wait()
    server.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
