"""kubelet_internal_checkpoint parser (reference
pkg/deviceplugin/checkpoint): maps allocated device IDs -> pod UID when
the pod cache is cold after a plugin restart.  Handles both the plain
and the NUMA-annotated checkpoint formats kubelet has shipped.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional

KUBELET_CHECKPOINT = \
    "/var/lib/kubelet/device-plugins/kubelet_internal_checkpoint"


def parse_kubelet_checkpoint(path: str = KUBELET_CHECKPOINT
                             ) -> Dict[str, dict]:
    """Returns {pod_uid: {container: {resource: [device_ids]}}}."""
    if not os.path.exists(path):
        return {}
    try:
        data = json.load(open(path))
    except (OSError, ValueError):
        return {}
    entries = (data.get("Data", {}) or {}).get("PodDeviceEntries") or []
    out: Dict[str, dict] = {}
    for e in entries:
        pod_uid = e.get("PodUID", "")
        cont = e.get("ContainerName", "")
        res = e.get("ResourceName", "")
        ids: List[str] = []
        dev = e.get("DeviceIDs")
        if isinstance(dev, list):
            # plain format: ["id1", "id2"]
            ids = [str(x) for x in dev]
        elif isinstance(dev, dict):
            # NUMA format: {"0": ["id1"], "1": ["id2"]}
            for v in dev.values():
                ids.extend(str(x) for x in v)
        out.setdefault(pod_uid, {}).setdefault(cont, {})[res] = ids
    return out


def pod_for_device(checkpoint: Dict[str, dict], resource: str,
                   device_id: str) -> Optional[tuple]:
    """(pod_uid, container) owning device_id under resource, or None."""
    for pod_uid, conts in checkpoint.items():
        for cont, resources in conts.items():
            if device_id in (resources.get(resource) or []):
                return pod_uid, cont
    return None
