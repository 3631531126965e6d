"""Cross-language ABI layout pin: C offsets/sizes vs Python ctypes.

Compiles library/test/abi_probe.c (via the library Makefile) and checks
every printed offsetof/sizeof against the ctypes mirrors — any drift in
either language fails here before it can corrupt a shared region.
"""
import ctypes
import os
import re
import subprocess

import pytest

from vgpu_manager_amd.config import abi

C2PY = {
    "region_header_t": abi.RegionHeader,
    "device_t": abi.DeviceT,
    "resource_data_t": abi.ResourceDataT,
    "pids_data_t": abi.PidsDataT,
    "util_proc_t": abi.UtilProcT,
    "device_util_t": abi.DeviceUtilT,
    "util_region_t": abi.UtilRegionT,
    "vmem_record_t": abi.VmemRecordT,
    "vmem_dev_counter_t": abi.VmemDevCounterT,
    "vmem_region_t": abi.VmemRegionT,
    "sm_node_dev_t": abi.SmNodeDevT,
    "sm_node_region_t": abi.SmNodeRegionT,
}

LINE_RE = re.compile(
    r"^(sizeof|offsetof)\((\w+)(?:,\s*(\w+))?\)=(\d+)$")


def test_abi_layout_matches_c(built_core):
    probe = os.path.join(built_core, "abi_probe")
    out = subprocess.run([probe], capture_output=True, text=True,
                         check=True).stdout
    checked = 0
    for line in out.strip().splitlines():
        m = LINE_RE.match(line.strip())
        assert m, f"unparseable probe line: {line!r}"
        kind, struct, fieldname, value = m.groups()
        value = int(value)
        cls = C2PY[struct]
        if kind == "sizeof":
            assert ctypes.sizeof(cls) == value, \
                f"sizeof({struct}): py={ctypes.sizeof(cls)} c={value}"
        else:
            fld = getattr(cls, fieldname)
            assert fld.offset == value, \
                f"offsetof({struct},{fieldname}): py={fld.offset} c={value}"
        checked += 1
    assert checked >= 60, f"probe only produced {checked} checks"


def test_magics_match_header():
    hook_h = os.path.join(os.path.dirname(__file__), "..", "library",
                          "include", "hook.h")
    src = open(hook_h).read()

    def c_macro(name):
        m = re.search(rf"#define\s+{name}\s+0x([0-9A-Fa-f]+)ULL", src)
        assert m, name
        return int(m.group(1), 16)

    assert abi.VGPU_CFG_MAGIC == c_macro("VGPU_CFG_MAGIC")
    assert abi.VGPU_UTIL_MAGIC == c_macro("VGPU_UTIL_MAGIC")
    assert abi.VGPU_VMEM_MAGIC == c_macro("VGPU_VMEM_MAGIC")
    assert abi.VGPU_SMND_MAGIC == c_macro("VGPU_SMND_MAGIC")
    assert abi.VGPU_PIDS_MAGIC == c_macro("VGPU_PIDS_MAGIC")
