"""DRA driver tests: prepare/unprepare idempotence, checkpoint
integrity + corruption handling, ResourceSlice shapes, CDI spec,
consumable shares, CPX partitions (reference pkg/kubeletplugin tests).
"""
import json
import os

import pytest

from vgpu_manager_amd.config.regions import VgpuConfigReader
from vgpu_manager_amd.device.types import fake_device
from vgpu_manager_amd.dra import cdi
from vgpu_manager_amd.dra.state import (
    CPX_PARTITIONS_PER_GPU,
    Checkpoint,
    DeviceState,
    VgpuClaimParams,
    build_resource_slice,
)


@pytest.fixture
def state(tmp_path):
    devices = [fake_device(i) for i in range(2)]
    return DeviceState(
        "node-a", devices,
        claims_dir=str(tmp_path / "claims"),
        checkpoint_path=str(tmp_path / "checkpoint.json"))


def test_prepare_writes_partition(state):
    prepared = state.prepare("claim-1", [
        VgpuClaimParams(uuid="GPU-fake-0000", cores=25,
                        memory_mib=4096)])
    assert prepared.cdi_device_ids == [
        cdi.qualified_name("GPU-fake-0000")]
    pdir = os.path.join(prepared.container_dir, "default")
    snap = VgpuConfigReader(
        os.path.join(pdir, "config", "vgpu.config")).snapshot()
    assert snap["devices"][0]["total_memory"] == 4096 << 20
    assert snap["devices"][0]["core_limit"] == 25
    edits = json.load(open(os.path.join(pdir, "edits.json")))
    assert any("VGPU_MEM_LIMIT_0=" in e for e in edits["env"])
    assert any(m["containerPath"] == "/etc/ld.so.preload"
               for m in edits["mounts"])


def test_prepare_idempotent(state):
    p1 = state.prepare("claim-1", [VgpuClaimParams(uuid="GPU-fake-0000")])
    p2 = state.prepare("claim-1", [VgpuClaimParams(uuid="GPU-fake-0001")])
    # second call returns the first result, does not re-prepare
    assert p1.cdi_device_ids == p2.cdi_device_ids


def test_multicontainer_partitions(state):
    prepared = state.prepare("claim-2", [
        VgpuClaimParams(uuid="GPU-fake-0000", partition_key="cont-a"),
        VgpuClaimParams(uuid="GPU-fake-0001", partition_key="cont-b"),
    ])
    assert os.path.isdir(os.path.join(prepared.container_dir, "cont-a"))
    assert os.path.isdir(os.path.join(prepared.container_dir, "cont-b"))


def test_unprepare_removes(state):
    p = state.prepare("claim-3", [VgpuClaimParams(uuid="GPU-fake-0000")])
    assert state.unprepare("claim-3")
    assert not os.path.exists(p.container_dir)
    assert not state.unprepare("claim-3")
    assert state.prepared_claims() == []


def test_checkpoint_survives_restart(state, tmp_path):
    state.prepare("claim-4", [VgpuClaimParams(uuid="GPU-fake-0000")])
    # new state from the same checkpoint: claim still prepared
    s2 = DeviceState("node-a", [fake_device(0), fake_device(1)],
                     claims_dir=str(tmp_path / "claims"),
                     checkpoint_path=str(tmp_path / "checkpoint.json"))
    assert s2.prepared_claims() == ["claim-4"]


def test_checkpoint_corruption_detected(tmp_path):
    path = str(tmp_path / "ckpt.json")
    c = Checkpoint(path)
    c.claims["x"] = {"cdi_device_ids": [], "container_dir": "/tmp/x"}
    c.save()
    raw = json.load(open(path))
    raw["payload"]["claims"]["evil"] = {}
    json.dump(raw, open(path, "w"))  # checksum now stale
    c2 = Checkpoint(path)
    assert c2.claims == {}  # rejected, empty start


def test_unknown_device_rejected(state):
    with pytest.raises(ValueError, match="unknown device"):
        state.prepare("claim-5", [VgpuClaimParams(uuid="GPU-nope")])


def test_resource_slice_gpu():
    devs = [fake_device(0), fake_device(1, healthy=False)]
    rs = build_resource_slice("node-a", devs)
    assert rs["spec"]["driver"] == "manager.amd.com"
    assert len(rs["spec"]["devices"]) == 2
    d0 = rs["spec"]["devices"][0]
    assert d0["basic"]["capacity"]["memory"]["value"] == "294912Mi"
    assert rs["spec"]["devices"][1]["basic"]["attributes"]["healthy"][
        "bool"] is False


def test_resource_slice_consumable_shares():
    rs = build_resource_slice("node-a", [fake_device(0, number=10)],
                              consumable_shares=True)
    cc = rs["spec"]["devices"][0]["basic"]["consumesCounters"][0]
    assert cc["counters"]["shares"]["value"] == "10"


def test_resource_slice_cpx():
    rs = build_resource_slice("node-a", [fake_device(0)], cpx=True)
    devs = rs["spec"]["devices"]
    assert len(devs) == CPX_PARTITIONS_PER_GPU
    assert devs[0]["basic"]["attributes"]["type"]["string"] == \
        "cpx-partition"
    # 288 GiB / 8 partitions
    assert devs[0]["basic"]["capacity"]["memory"]["value"] == \
        f"{294912 // 8}Mi"


def test_cdi_spec_write(tmp_path):
    devs = [fake_device(0)]
    spec = cdi.build_cdi_spec(devs)
    path = cdi.write_cdi_spec(spec, cdi_dir=str(tmp_path))
    data = json.load(open(path))
    assert data["kind"] == "k8s.device-plugin.amd.com/gpu"
    nodes = data["devices"][0]["containerEdits"]["deviceNodes"]
    assert {n["path"] for n in nodes} == {"/dev/kfd",
                                          "/dev/dri/renderD128"}
