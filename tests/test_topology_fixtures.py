"""Topology fixture tests on realistic hardware shapes (reference
pkg/device/allocator/upstream_fixtures_test.go replays real
`nvidia-smi topo -m` matrices; these replay MI355X-era shapes):

  * the standard 8-GPU OAM baseboard: fully-connected xGMI mesh,
    2 NUMA domains of 4;
  * a PCIe-only box (no xGMI): NUMA-local PIX links, cross-NUMA SYS;
  * a degraded mesh (one dead link) that forces real link scoring.
"""
import pytest

from vgpu_manager_amd.device.allocator import (
    AllocationError,
    Allocator,
    build_allocation_request,
)
from vgpu_manager_amd.device.types import (
    LINK_SYS,
    fake_node,
)
from vgpu_manager_amd.util import consts

from tests.test_allocator import make_pod


def alloc(node, pod):
    return Allocator(node).allocate(build_allocation_request(pod))


def claimed_ids(cdcs):
    return [c.id for c in cdcs[0].claims]


def numa_of(node, dev_id):
    return node.devices[dev_id].info.numa


# ---- fixture 1: standard OAM board (full xGMI, 2 NUMA x 4) ----

def test_oam_2gpu_numa_mode_stays_on_socket():
    node = fake_node("oam", 8)
    pod = make_pod(number=2,
                   ann={consts.topology_mode_ann(): consts.TOPO_NUMA})
    ids = claimed_ids(alloc(node, pod))
    assert len(ids) == 2
    assert numa_of(node, ids[0]) == numa_of(node, ids[1])


def test_oam_6gpu_numa_strict_fails_but_numa_ok():
    # 6 GPUs cannot fit one 4-GPU NUMA domain
    node = fake_node("oam", 8)
    strict = make_pod(number=6, ann={
        consts.topology_mode_ann(): consts.TOPO_NUMA_STRICT})
    with pytest.raises(AllocationError):
        alloc(node, strict)
    soft = make_pod(number=6, ann={
        consts.topology_mode_ann(): consts.TOPO_NUMA})
    assert len(claimed_ids(alloc(node, soft))) == 6


def test_oam_link_mode_uniform_mesh_any_pair_ok():
    # on a full mesh every pair has identical cost: link mode must
    # succeed and still return a valid pair
    node = fake_node("oam", 8)
    pod = make_pod(number=2, ann={
        consts.topology_mode_ann(): consts.TOPO_LINK_STRICT})
    ids = claimed_ids(alloc(node, pod))
    assert len(ids) == 2


# ---- fixture 2: PCIe-only box (no xGMI) ----

def test_pcie_box_link_mode_prefers_same_numa():
    node = fake_node("pcie", 8, full_xgmi=False)
    pod = make_pod(number=4, ann={
        consts.topology_mode_ann(): consts.TOPO_LINK})
    ids = claimed_ids(alloc(node, pod))
    numas = {numa_of(node, i) for i in ids}
    assert len(numas) == 1, f"4-GPU set crossed NUMA: {ids}"


def test_pcie_box_link_strict_5gpu_must_cross():
    # 5 GPUs cannot avoid a SYS link; strict still succeeds (there is
    # no tier below SYS) but the chosen set minimizes SYS pairs:
    # 4 + 1 split has 4 cross pairs; any other split has more
    node = fake_node("pcie", 8, full_xgmi=False)
    pod = make_pod(number=5, ann={
        consts.topology_mode_ann(): consts.TOPO_LINK})
    ids = claimed_ids(alloc(node, pod))
    from collections import Counter
    counts = Counter(numa_of(node, i) for i in ids)
    assert sorted(counts.values()) == [1, 4]


# ---- fixture 3: degraded mesh (one link down) ----

def test_degraded_mesh_avoids_broken_link():
    node = fake_node("degraded", 8)
    # sever xGMI 0<->1 in both directions (link becomes SYS)
    node.topology.devices[0].links[1].kind = LINK_SYS
    node.topology.devices[1].links[0].kind = LINK_SYS
    pod = make_pod(number=2, ann={
        consts.topology_mode_ann(): consts.TOPO_LINK})
    # run several times over fresh nodes to make sure {0,1} is never
    # chosen while 27 other perfect pairs exist
    for _ in range(5):
        ids = sorted(claimed_ids(alloc(node, pod)))
        assert ids != [0, 1], "allocator picked the severed pair"
