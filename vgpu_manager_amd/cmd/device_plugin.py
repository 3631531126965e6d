"""device-plugin binary (reference cmd/device-plugin): node config,
device manager, plugin set, registry server (client mode), reschedule
controller, kubelet-restart watch, utilization sampler."""
from __future__ import annotations

import argparse
import logging
import os
import signal
import sys
import threading

from ..client.kube import RestKubeClient
from ..controller.reschedule import RescheduleController
from ..device.manager import AmdDeviceManager
from ..deviceplugin import api
from ..deviceplugin.server import PluginSet, watch_kubelet_restart
from ..registry.server import RegistryServer, RegistryState
from ..util import consts
from ..util.nodeconfig import CORE_GATES, FeatureGates, load_node_config


def main(argv=None):
    ap = argparse.ArgumentParser("vgpu-device-plugin")
    ap.add_argument("--node-name",
                    default=os.environ.get("NODE_NAME", ""))
    ap.add_argument("--node-config-path", default=None)
    ap.add_argument("--feature-gates", default="")
    ap.add_argument("--device-split-count", type=int, default=None)
    ap.add_argument("--device-memory-scaling", type=float, default=None)
    ap.add_argument("--domain", default=consts.AMD_DOMAIN)
    ap.add_argument("--kubelet-socket", default=api.KUBELET_SOCKET)
    ap.add_argument("--driver-lib",
                    default="/usr/local/vgpu-manager/"
                            + consts.DRIVER_LIB_NAME)
    args = ap.parse_args(argv)

    logging.basicConfig(level=logging.INFO)
    consts.set_domain(args.domain)
    if not args.node_name:
        ap.error("--node-name or NODE_NAME required")

    gates = FeatureGates(CORE_GATES)
    gates.parse(args.feature_gates)
    gates.validate(gates.as_dict())

    config = load_node_config(args.node_config_path, args.node_name)
    if args.device_split_count is not None:
        config.deviceSplitCount = args.device_split_count
    if args.device_memory_scaling is not None:
        config.deviceMemoryScaling = args.device_memory_scaling

    client = RestKubeClient()
    manager = AmdDeviceManager(args.node_name, config)
    manager.register(client)
    manager.heartbeat_loop(client)

    plugin_set = PluginSet(
        manager, client, open_vcore=config.openVCore,
        open_vmemory=config.openVMemory, driver_lib=args.driver_lib,
        shared_watcher=gates.enabled("SharedSMUtilizationWatcher"),
        client_mode=gates.enabled("DevicePluginClientMode"))
    plugin_set.start_all(args.kubelet_socket)
    watch_kubelet_restart(plugin_set, args.kubelet_socket)

    if gates.enabled("DevicePluginClientMode"):
        state = RegistryState()
        reg = RegistryServer(
            os.path.join(consts.MANAGER_DIR, "registry", "socket.sock"),
            state)
        reg.start_background()

    if gates.enabled("SharedSMUtilizationWatcher"):
        from ..monitor.sampler import AmdSmiSource, UtilSampler
        sampler = UtilSampler(
            AmdSmiSource(),
            os.path.join(consts.MANAGER_DIR, "watcher",
                         "sm_util.config"))
        sampler.start_background()

    if gates.enabled("AllocationFailureReschedule"):
        ctl = RescheduleController(
            client, args.node_name,
            os.path.join(consts.MANAGER_DIR, "reschedule-checkpoint.json"))
        threading.Thread(target=ctl.run_forever, daemon=True).start()

    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    signal.signal(signal.SIGINT, lambda *_: stop.set())
    stop.wait()
    plugin_set.stop_all()
    return 0


if __name__ == "__main__":
    sys.exit(main())
