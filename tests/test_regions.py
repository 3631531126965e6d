"""Python-side region writer/reader round-trips + runtime mutation."""
import os

import pytest

from vgpu_manager_amd.config.abi import (
    DEV_FLAG_CORE_LIMIT,
    DEV_FLAG_MEM_LIMIT,
    DEV_FLAG_OVERSOLD,
)
from vgpu_manager_amd.config.regions import (
    DeviceLimit,
    PidsWriter,
    UtilRegionWriter,
    VgpuConfigReader,
    VgpuConfigWriter,
)


def test_vgpu_config_roundtrip(tmp_path):
    path = str(tmp_path / "config" / "vgpu.config")
    w = VgpuConfigWriter(path)
    w.write(
        pod_uid="uid-123", pod_name="p", pod_namespace="ns",
        container_name="main",
        limits=[
            DeviceLimit(uuid="GPU-aaaa", host_index=3,
                        memory_bytes=1 << 30, core_limit=25,
                        oversold=True),
            DeviceLimit(uuid="GPU-bbbb", host_index=5),
        ],
        compute_policy="balance", oversold=False)

    r = VgpuConfigReader(path)
    snap = r.snapshot()
    assert snap["pod_uid"] == "uid-123"
    assert snap["container_name"] == "main"
    assert len(snap["devices"]) == 2
    d0, d1 = snap["devices"]
    assert d0["total_memory"] == 1 << 30
    assert d0["core_limit"] == 25
    assert d0["host_index"] == 3
    assert d0["flags"] & DEV_FLAG_MEM_LIMIT
    assert d0["flags"] & DEV_FLAG_CORE_LIMIT
    assert d0["flags"] & DEV_FLAG_OVERSOLD
    assert d1["total_memory"] == 0
    assert not (d1["flags"] & DEV_FLAG_MEM_LIMIT)
    assert d1["uuid"].rstrip("\x00") == "GPU-bbbb"
    r.close()
    w.close()


def test_vgpu_config_runtime_modify(tmp_path):
    path = str(tmp_path / "vgpu.config")
    w = VgpuConfigWriter(path)
    w.write(pod_uid="u", pod_name="p", pod_namespace="n",
            container_name="c",
            limits=[DeviceLimit(uuid="GPU-x", host_index=0,
                                memory_bytes=100, core_limit=10)])
    w.modify_device(0, memory_bytes=200, core_limit=50)
    r = VgpuConfigReader(path)
    d = r.snapshot()["devices"][0]
    assert d["total_memory"] == 200
    assert d["core_limit"] == 50
    # seqlock seq must be even after two writes
    assert w.region.data.devices[0].seq % 2 == 0
    assert w.region.data.devices[0].seq >= 4
    r.close()
    w.close()


def test_vgpu_config_reader_rejects_garbage(tmp_path):
    path = str(tmp_path / "garbage")
    with open(path, "wb") as f:
        f.write(b"\x00" * 100)
    with pytest.raises(ValueError):
        VgpuConfigReader(path)


def test_pids_writer(tmp_path):
    path = str(tmp_path / "pids.config")
    w = PidsWriter(path)
    w.write([30, 10, 20, 10])
    d = w.region.data
    assert d.pid_count == 3
    assert list(d.pids[:3]) == [10, 20, 30]
    w.close()


def test_util_writer(tmp_path):
    path = str(tmp_path / "sm_util.config")
    w = UtilRegionWriter(path, device_count=2)
    w.publish(0, dev_busy_permille=750, vram_used_bytes=123456,
              procs=[dict(pid=42, gfx_busy_permille=500,
                          vram_bytes=1000, cu_occupancy=64)])
    d = w.region.data.devices[0]
    assert d.seq % 2 == 0 and d.seq >= 2
    assert d.dev_busy_permille == 750
    assert d.proc_count == 1
    assert d.procs[0].pid == 42
    assert w.region.data.heartbeat_ns > 0
    w.close()
