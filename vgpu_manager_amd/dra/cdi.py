"""CDI spec generation (reference pkg/deviceplugin/cdi + the DRA
driver's CDI edits).  Vendor `k8s.device-plugin.amd.com`, class `gpu`;
per-UUID devices with containerEdits injecting /dev/kfd + the GPU's
renderD node; the vGPU partition edits add the shim mounts + envs.
"""
from __future__ import annotations

import json
import os
from typing import Dict, Optional

from ..util import consts

CDI_VERSION = "0.6.0"
CDI_VENDOR = "k8s.device-plugin.amd.com"
CDI_CLASS = "gpu"
CDI_KIND = f"{CDI_VENDOR}/{CDI_CLASS}"
CDI_DIR = "/etc/cdi"


def device_node_edits(host_index: int) -> dict:
    return {
        "deviceNodes": [
            {"path": "/dev/kfd", "permissions": "rw"},
            {"path": f"/dev/dri/renderD{128 + host_index}",
             "permissions": "rw"},
        ]
    }


def vgpu_container_edits(*, driver_lib: str, container_dir: str,
                         envs: Dict[str, str],
                         shared_watcher_dir: Optional[str] = None) -> dict:
    """The per-claim container edits equivalent to the device plugin's
    Allocate response (reference kubeletplugin/vgpu.go:185-436)."""
    mounts = [
        {"hostPath": driver_lib,
         "containerPath":
             f"{consts.MANAGER_DIR}/driver/{consts.DRIVER_LIB_NAME}",
         "options": ["ro", "bind"]},
        {"hostPath": os.path.join(container_dir, "ld.so.preload"),
         "containerPath": "/etc/ld.so.preload",
         "options": ["ro", "bind"]},
        {"hostPath": os.path.join(container_dir, "config"),
         "containerPath": f"{consts.MANAGER_DIR}/config",
         "options": ["ro", "bind"]},
        {"hostPath": os.path.join(container_dir, "vgpu_lock"),
         "containerPath": "/tmp/.vgpu_lock",
         "options": ["rw", "bind"]},
        {"hostPath": os.path.join(container_dir, "vmem_node"),
         "containerPath": "/tmp/.vmem_node",
         "options": ["rw", "bind"]},
        {"hostPath": os.path.join(container_dir, "sm_node"),
         "containerPath": "/tmp/.sm_node",
         "options": ["rw", "bind"]},
    ]
    if shared_watcher_dir:
        mounts.append({"hostPath": shared_watcher_dir,
                       "containerPath": f"{consts.MANAGER_DIR}/watcher",
                       "options": ["ro", "bind"]})
    return {
        "env": [f"{k}={v}" for k, v in sorted(envs.items())],
        "mounts": mounts,
    }


def build_cdi_spec(devices: list, *, spec_devices: Optional[list] = None
                   ) -> dict:
    """devices: DeviceInfo list -> a CDI spec dict."""
    cdi_devices = spec_devices or []
    if not cdi_devices:
        for d in devices:
            cdi_devices.append({
                "name": d.uuid,
                "containerEdits": device_node_edits(d.id),
            })
    return {
        "cdiVersion": CDI_VERSION,
        "kind": CDI_KIND,
        "devices": cdi_devices,
    }


def write_cdi_spec(spec: dict, cdi_dir: str = CDI_DIR) -> str:
    os.makedirs(cdi_dir, exist_ok=True)
    path = os.path.join(cdi_dir, f"{CDI_VENDOR}-{CDI_CLASS}.json")
    tmp = path + ".tmp"
    with open(tmp, "w") as f:
        json.dump(spec, f, indent=2)
    os.replace(tmp, path)
    return path


def qualified_name(device_name: str) -> str:
    return f"{CDI_KIND}={device_name}"
