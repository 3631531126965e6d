#!/usr/bin/env python3
"""Node preflight: verify a (would-be) vGPU node satisfies every
assumption the stack depends on, before deploying the DaemonSets.

Each check maps to a failure mode documented in
docs/troubleshooting.md; run with --json for machine output.  All
filesystem roots are overridable so the checker itself is testable
on CPU-only boxes (override envs follow the shim's *_OVERRIDE
naming convention; `VGPU_PROC_DIR_OVERRIDE` is shared with the shim).

Usage:  python scripts/preflight.py [--json] [--strict]
"""
from __future__ import annotations

import argparse
import ctypes
import ctypes.util
import json
import os
import sys

OK, WARN, FAIL = "ok", "warn", "fail"


def _root(env, default):
    return os.environ.get(env, default)


def check_kfd_nodes(results):
    """KFD topology present → amdgpu driver loaded, GPUs visible."""
    root = _root("VGPU_KFD_TOPO_OVERRIDE",
                 "/sys/class/kfd/kfd/topology/nodes")
    gpus = 0
    if os.path.isdir(root):
        for n in sorted(os.listdir(root)):
            props = os.path.join(root, n, "properties")
            try:
                txt = open(props).read()
            except OSError:
                continue
            if "simd_count" in txt and not any(
                    line.split() == ["simd_count", "0"]
                    for line in txt.splitlines()):
                gpus += 1
    results.append(("kfd_gpu_nodes", OK if gpus else FAIL,
                    f"{gpus} GPU node(s) in KFD topology"))
    return gpus


def check_device_nodes(results):
    """/dev/kfd + renderD* are what the plugin mounts into pods."""
    kfd = _root("VGPU_DEV_KFD_OVERRIDE", "/dev/kfd")
    dri = _root("VGPU_DEV_DRI_OVERRIDE", "/dev/dri")
    have_kfd = os.path.exists(kfd)
    renders = [f for f in (os.listdir(dri) if os.path.isdir(dri) else [])
               if f.startswith("renderD")]
    results.append(("dev_kfd", OK if have_kfd else FAIL, kfd))
    results.append(("dev_dri_render", OK if renders else FAIL,
                    f"{len(renders)} render node(s)"))


def check_amdsmi(results):
    """The node-side watcher and `max` memory accounting use
    libamd_smi; in-container attribution no longer requires it
    (shim ladder source 4) so host absence is WARN, not FAIL."""
    name = os.environ.get("VGPU_AMDSMI_PATH") or \
        ctypes.util.find_library("amd_smi") or "libamd_smi.so"
    try:
        ctypes.CDLL(name)
        results.append(("libamd_smi", OK, name))
    except OSError as e:
        results.append(("libamd_smi", WARN,
                        f"not loadable ({e}); node watcher degrades "
                        "to KFD occupancy, max-mode accounting to "
                        "ledger"))


def check_lock_dirs(results):
    """Cross-process locks and shared regions live under /tmp —
    must be writable from the agent's mount namespace."""
    for env, d in (("VGPU_LOCK_DIR_OVERRIDE", "/tmp/.vgpu_lock"),
                   ("VGPU_VMEM_DIR_OVERRIDE", "/tmp/.vmem_node"),
                   ("VGPU_SMND_DIR_OVERRIDE", "/tmp/.sm_node")):
        path = _root(env, d)
        parent = os.path.dirname(path) or "/"
        writable = os.access(parent, os.W_OK)
        results.append((f"writable:{os.path.basename(path)}",
                        OK if writable else FAIL, parent))


def check_xnack(results):
    """Managed-memory spill (VGPU_MEM_OVERSOLD) prefers HSA XNACK;
    without it the shim falls back to pinned-host mapping."""
    root = _root("VGPU_KFD_TOPO_OVERRIDE",
                 "/sys/class/kfd/kfd/topology/nodes")
    found = None
    if os.path.isdir(root):
        for n in sorted(os.listdir(root)):
            try:
                txt = open(os.path.join(root, n, "properties")).read()
            except OSError:
                continue
            for line in txt.splitlines():
                parts = line.split()
                if parts[:1] == ["capability"] and len(parts) == 2:
                    # bit 2 of HSA capability word = XNACK supported
                    found = bool(int(parts[1]) & 0x4) or found
    if found is None:
        results.append(("xnack", WARN, "capability not readable"))
    else:
        results.append(("xnack", OK if found else WARN,
                        "supported" if found else
                        "absent: oversold spill uses pinned host "
                        "memory (slower)"))


def check_host_proc(results):
    """Attribution without client-mode needs the host /proc mounted
    at .host_proc in the plugin container (pod-UID cgroup walk)."""
    hp = _root("VGPU_PROC_DIR_OVERRIDE", "/.host_proc")
    if os.path.isdir(hp):
        results.append(("host_proc", OK, hp))
    elif os.path.samefile("/proc/self", "/proc/%d" % os.getpid()):
        results.append(("host_proc", WARN,
                        f"{hp} absent — fine if running on the host "
                        "or in DevicePluginClientMode"))


def check_ipc_env(results):
    """Multi-process RCCL / torch CUDA-tensor sharing on this driver
    stack needs dmabuf IPC."""
    v = os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY")
    results.append(("HSA_ENABLE_IPC_MODE_LEGACY",
                    OK if v == "0" else WARN,
                    f"={v!r} (want '0'; the plugin injects it into "
                    "pods, export it for host-side tooling)"))


def main(argv=None):
    ap = argparse.ArgumentParser("vgpu-preflight")
    ap.add_argument("--json", action="store_true")
    ap.add_argument("--strict", action="store_true",
                    help="treat warnings as failures")
    args = ap.parse_args(argv)

    results = []
    check_kfd_nodes(results)
    check_device_nodes(results)
    check_amdsmi(results)
    check_lock_dirs(results)
    check_xnack(results)
    check_host_proc(results)
    check_ipc_env(results)

    bad = {FAIL, WARN} if args.strict else {FAIL}
    rc = 1 if any(s in bad for _, s, _ in results) else 0
    if args.json:
        print(json.dumps([{"check": c, "status": s, "detail": d}
                          for c, s, d in results]))
    else:
        for c, s, d in results:
            print(f"[{s.upper():4}] {c}: {d}")
        print("preflight:", "FAIL" if rc else "PASS")
    return rc


if __name__ == "__main__":
    sys.exit(main())
