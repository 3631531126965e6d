"""Per-node YAML config + feature gates (reference pkg/config/node +
the two gate registries).

Node config file overrides CLI defaults per node (deviceSplitCount,
memory/cores scaling, exclude devices, cgroup driver, open vcore/vmem).
The persistent fake-device-ID store keeps kubelet-visible device IDs
stable across plugin restarts (reference id_store.go).
"""
from __future__ import annotations

import json
import os
import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import yaml

from ..device.types import NodeConfigInfo

# ---- feature gates (reference feature_gates.md: two registries) ----

CORE_GATES = {
    "SerializedNodeFilter": True,
    "SerializedNodeBind": True,
    "AllocationFailureReschedule": True,
    "SharedSMUtilizationWatcher": False,
    "DevicePluginClientMode": False,
    "GPUTopologyAwareScheduling": True,
    "VirtualMemoryOversold": False,
}

DRA_GATES = {
    "DRADriver": False,
    "ConsumableShares": False,
    "DynamicCPXPartitioning": False,
    "VFIOPassthrough": False,
    "NRIMountIsolation": False,
}

# gate dependency / mutual-exclusion rules checked at startup
GATE_RULES = [
    # (gate, requires, conflicts)
    ("ConsumableShares", ["DRADriver"], []),
    ("DynamicCPXPartitioning", ["DRADriver"], []),
    ("NRIMountIsolation", ["DRADriver"], []),
    ("DRADriver", [], ["DevicePluginClientMode"]),
]


class FeatureGates:
    def __init__(self, defaults: Dict[str, bool]):
        self._gates = dict(defaults)

    def parse(self, spec: str) -> None:
        """'Gate1=true,Gate2=false' CLI form."""
        for part in (spec or "").split(","):
            part = part.strip()
            if not part:
                continue
            name, _, val = part.partition("=")
            if name not in self._gates:
                raise ValueError(f"unknown feature gate {name!r}")
            self._gates[name] = val.lower() in ("1", "true", "on", "")

    def enabled(self, name: str) -> bool:
        return bool(self._gates.get(name, False))

    def validate(self, all_gates: Dict[str, bool]) -> None:
        for gate, requires, conflicts in GATE_RULES:
            if not all_gates.get(gate):
                continue
            for r in requires:
                if not all_gates.get(r):
                    raise ValueError(f"gate {gate} requires {r}")
            for c in conflicts:
                if all_gates.get(c):
                    raise ValueError(f"gate {gate} conflicts with {c}")

    def as_dict(self) -> Dict[str, bool]:
        return dict(self._gates)


def load_node_config(path: Optional[str], node_name: str
                     ) -> NodeConfigInfo:
    """YAML file with global defaults + per-node overrides:

        deviceSplitCount: 10
        deviceMemoryScaling: 1.0
        nodes:
          node-a:
            deviceSplitCount: 4
            excludeDevices: [7]
    """
    cfg = NodeConfigInfo()
    if not path or not os.path.exists(path):
        return cfg
    data = yaml.safe_load(open(path)) or {}
    merged = {k: v for k, v in data.items() if k != "nodes"}
    per_node = (data.get("nodes") or {}).get(node_name) or {}
    merged.update(per_node)
    known = set(NodeConfigInfo.__dataclass_fields__)
    for k, v in merged.items():
        if k in known:
            setattr(cfg, k, v)
    return cfg


class FakeIdStore:
    """Persistent fake-device-ID order so kubelet device IDs survive
    plugin restarts (reference pkg/config/node/id_store.go)."""

    def __init__(self, path: str):
        self.path = path
        self._mu = threading.Lock()
        try:
            self.order: List[str] = json.load(open(path))
        except (OSError, ValueError):
            self.order = []

    def stable_order(self, uuids: List[str]) -> List[str]:
        with self._mu:
            kept = [u for u in self.order if u in uuids]
            new = [u for u in uuids if u not in kept]
            self.order = kept + sorted(new)
            tmp = self.path + ".tmp"
            os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
            with open(tmp, "w") as f:
                json.dump(self.order, f)
            os.replace(tmp, self.path)
            return list(self.order)
