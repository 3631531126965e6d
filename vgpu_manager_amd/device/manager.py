"""Node device manager: discovery (amd-smi), health, registration.

Reference pkg/device/manager re-designed for MI355X: discovery via the
amdsmi python binding (serial/uuid, VRAM, CU count, NUMA via sysfs,
xGMI links via amdsmi topo), node-annotation registration + heartbeat,
and a health loop that marks devices unhealthy on RAS/query failures.
`FakeDeviceManager` backs every CPU test (reference NewFakeDeviceManager
pattern).
"""
from __future__ import annotations

import logging
import os
import threading
import time
from typing import Callable, List, Optional

from ..client.kube import KubeClient
from ..util import consts
from .types import (
    DeviceInfo,
    DeviceLink,
    DeviceTopology,
    LINK_PCIE_NUMA,
    LINK_SYS,
    LINK_XGMI,
    NodeConfigInfo,
    NodeTopologyInfo,
    encode_node_devices,
)

log = logging.getLogger("vgpu.device.manager")


class DeviceManager:
    """Base: holds devices + topology and handles node registration."""

    def __init__(self, node_name: str, config: Optional[NodeConfigInfo] =
                 None):
        self.node_name = node_name
        self.config = config or NodeConfigInfo()
        self.devices: List[DeviceInfo] = []
        self.topology = NodeTopologyInfo()
        self._health_cbs: List[Callable[[DeviceInfo], None]] = []
        self._stop = threading.Event()

    # ---- registration (annotations) ----
    def register(self, client: KubeClient) -> None:
        anns = {
            consts.node_register_ann(): encode_node_devices(self.devices),
            consts.node_topology_ann(): self.topology.encode(),
            consts.node_config_ann(): self.config.encode(),
            consts.node_heartbeat_ann(): str(int(time.time())),
        }
        client.patch_node_annotations(self.node_name, anns)

    def heartbeat_loop(self, client: KubeClient,
                       interval_s: int = 30) -> threading.Thread:
        def run():
            while not self._stop.wait(interval_s):
                try:
                    self.refresh_health()
                    self.register(client)
                except Exception as e:  # keep beating
                    log.warning("heartbeat failed: %s", e)
        t = threading.Thread(target=run, daemon=True,
                             name="vgpu-heartbeat")
        t.start()
        return t

    def stop(self) -> None:
        self._stop.set()

    def on_health_change(self, cb: Callable[[DeviceInfo], None]) -> None:
        self._health_cbs.append(cb)

    def refresh_health(self) -> None:
        pass

    def _notify(self, dev: DeviceInfo) -> None:
        for cb in self._health_cbs:
            cb(dev)


class FakeDeviceManager(DeviceManager):
    def __init__(self, node_name: str = "fake-node", n_devices: int = 8,
                 config: Optional[NodeConfigInfo] = None, **dev_kwargs):
        super().__init__(node_name, config)
        from .types import fake_node
        info = fake_node(node_name, n_devices, **dev_kwargs)
        self.devices = [u.info for u in info.devices.values()]
        for d in self.devices:
            d.number = self.config.deviceSplitCount
        self.topology = info.topology

    def set_health(self, dev_id: int, healthy: bool) -> None:
        for d in self.devices:
            if d.id == dev_id and d.healthy != healthy:
                d.healthy = healthy
                self._notify(d)


class AmdDeviceManager(DeviceManager):
    """Real discovery through the amdsmi python binding."""

    def __init__(self, node_name: str,
                 config: Optional[NodeConfigInfo] = None):
        super().__init__(node_name, config)
        import amdsmi
        self.amdsmi = amdsmi
        amdsmi.amdsmi_init()
        self._handles = amdsmi.amdsmi_get_processor_handles()
        self.discover()

    def discover(self) -> None:
        a = self.amdsmi
        devices, topo = [], NodeTopologyInfo()
        split = self.config.deviceSplitCount
        for i, h in enumerate(self._handles):
            uuid, mem_mib, name, numa, bus = f"GPU-{i}", 294912, \
                "MI355X", -1, ""
            try:
                uuid = str(a.amdsmi_get_gpu_device_uuid(h))
            except Exception:
                pass
            try:
                asic = a.amdsmi_get_gpu_asic_info(h)
                name = asic.get("market_name") or name
            except Exception:
                pass
            try:
                mem = a.amdsmi_get_gpu_memory_total(
                    h, a.AmdSmiMemoryType.VRAM)
                mem_mib = int(mem) >> 20
            except Exception:
                pass
            try:
                bdf = a.amdsmi_get_gpu_device_bdf(h)
                bus = str(bdf)
                numa_path = f"/sys/bus/pci/devices/{bus.lower()}" \
                            "/numa_node"
                if os.path.exists(numa_path):
                    numa = int(open(numa_path).read().strip())
            except Exception:
                pass
            scaled_mem = int(mem_mib * self.config.deviceMemoryScaling)
            devices.append(DeviceInfo(
                id=i, type=name, uuid=uuid, core=100, memory=scaled_mem,
                number=split, numa=numa, busId=bus, healthy=True))
            topo.devices.append(DeviceTopology(id=i, uuid=uuid, numa=numa))

        # xGMI link topology
        for i, hi in enumerate(self._handles):
            for j, hj in enumerate(self._handles):
                if i == j:
                    continue
                kind, weight, hops = LINK_SYS, 100, 2
                try:
                    lt = self.amdsmi.amdsmi_topo_get_link_type(hi, hj)
                    # dict with 'hops' and 'type' on current bindings
                    t = lt.get("type") if isinstance(lt, dict) else lt
                    hops = lt.get("hops", 1) if isinstance(lt, dict) else 1
                    tname = str(t).upper()
                    if "XGMI" in tname:
                        kind = LINK_XGMI
                    elif "PCIE" in tname:
                        same_numa = (topo.devices[i].numa ==
                                     topo.devices[j].numa)
                        kind = LINK_PCIE_NUMA if same_numa else LINK_SYS
                except Exception:
                    pass
                try:
                    w = self.amdsmi.amdsmi_topo_get_link_weight(hi, hj)
                    weight = int(w)
                except Exception:
                    pass
                topo.devices[i].links[j] = DeviceLink(
                    peer_id=j, kind=kind, weight=weight, hops=hops)

        if self.config.excludeDevices:
            devices = [d for d in devices
                       if d.id not in self.config.excludeDevices]
        self.devices = devices
        self.topology = topo

    def refresh_health(self) -> None:
        a = self.amdsmi
        for i, h in enumerate(self._handles):
            healthy = True
            try:
                a.amdsmi_get_gpu_activity(h)
                try:
                    ras = a.amdsmi_get_gpu_total_ecc_count(h)
                    if isinstance(ras, dict) and \
                            ras.get("uncorrectable_count", 0) > 0:
                        healthy = False
                except Exception:
                    pass
            except Exception:
                healthy = False
            for d in self.devices:
                if d.id == i and d.healthy != healthy:
                    d.healthy = healthy
                    self._notify(d)
